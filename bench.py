"""Flagship benchmark: GBMClassifier training throughput (BASELINE.json).

Measures rows/sec (whole node) for GBMClassifier boosting rounds with
depth-8 histogram trees on synthetic 10M x 256 fp32 tabular data, 10M rows
PER GPU (weak scaling; BASELINE config 2 at N=1, config 3 shape at N=8).

One step == one boosting round: fused gradient computation, one depth-8
tree fit (per-level LDS histogram build + RCCL histogram all-reduce +
split search + partition), Brent line search of the stage weight, and the
margin update over all rows.

Contract (driver): `python bench.py --gpus N --steps K --warmup W`; for
N > 1 the driver launches one rank per GPU via torch.distributed.run; rank 0
prints ONE JSON line.
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time

import torch


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=8)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--rows", type=int, default=10_000_000, help="rows per GPU")
    ap.add_argument("--features", type=int, default=256)
    ap.add_argument("--max-depth", type=int, default=8)
    ap.add_argument("--max-bins", type=int, default=256)
    ap.add_argument("--trees", type=int, default=100, help="config tree count")
    ap.add_argument("--cpu", action="store_true", help="debug on CPU")
    args = ap.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))

    from spark_ensemble_amd.parallel import Comm, get_comm, init_from_env, set_comm

    if world > 1:
        comm = init_from_env()
    else:
        comm = Comm()
        set_comm(comm)

    if args.cpu or not torch.cuda.is_available():
        device = torch.device("cpu")
    else:
        torch.cuda.set_device(local_rank)
        device = torch.device("cuda", local_rank)

    import spark_ensemble_amd as sea
    from spark_ensemble_amd.boosting.losses import get_classification_loss
    from spark_ensemble_amd.boosting.line_search import optimize_weight_1d
    from spark_ensemble_amd.ensemble.binning import BinnedDataset
    from spark_ensemble_amd.models import DecisionTreeRegressor
    from spark_ensemble_amd.utils.io import synthetic_classification

    n = args.rows
    f = args.features

    # ---- data: synthetic binary classification, rank-sharded -------------
    t0 = time.time()
    df = synthetic_classification(
        n * world, f, k=2, seed=1234, device=device, shard=(rank, world),
        informative=48,
    )
    x, y = df["features"], df["label"]
    if rank == 0:
        print(f"# data ready in {time.time()-t0:.1f}s: {tuple(x.shape)} on {device}",
              file=sys.stderr)

    # ---- GBM setup (the inner loop of GBMClassifier._fit, exposed so the
    #      driver can time exactly K rounds) ------------------------------
    loss = get_classification_loss("bernoulli")  # dim=1: one tree per round
    learner = (
        DecisionTreeRegressor()
        .setMaxDepth(args.max_depth)
        .setMaxBins(args.max_bins)
    )
    gbm = sea.GBMClassifier()  # for fit_base_learner plumbing

    binned = BinnedDataset(x, df)
    t0 = time.time()
    binned.get(args.max_bins)  # bin once (setup, like the reference's persist)
    if device.type == "cuda":
        torch.cuda.synchronize()
    if rank == 0:
        print(f"# binning in {time.time()-t0:.1f}s", file=sys.stderr)

    ylab = loss.encode_label(y)
    import math

    p1 = float(y.mean())
    logodds = math.log(p1 / max(1e-12, 1.0 - p1))
    predictions = torch.full((x.shape[0], 1), logodds, device=device)

    def one_round(i: int):
        nonlocal predictions
        res_label = -loss.grad_hess_fused(ylab, predictions)[0].squeeze(1)
        fr = binned.fit_frame(learner, res_label)
        model = gbm.fit_base_learner(learner, fr)
        direction = getattr(model, "_train_pred", None)
        if direction is None:
            direction = model.predict(x)
        direction = direction.unsqueeze(1)
        w = torch.ones(x.shape[0], device=device)
        alpha = optimize_weight_1d(
            loss, ylab, predictions, direction, w, comm, max_iter=100, tol=1e-6
        )
        predictions = predictions + alpha * direction

    # ---- warmup ----------------------------------------------------------
    for i in range(args.warmup):
        one_round(i)

    # ---- timed region ----------------------------------------------------
    comm.barrier()
    if device.type == "cuda":
        torch.cuda.synchronize()
    t_start = time.time()
    for i in range(args.steps):
        one_round(args.warmup + i)
    if device.type == "cuda":
        torch.cuda.synchronize()
    comm.barrier()
    elapsed = time.time() - t_start
    # MAX over ranks
    elapsed = comm.all_reduce_scalar(elapsed, "max")

    global_rows = n * world
    value = global_rows * args.steps / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        out = {
            "metric": "rows/sec (whole node) GBMClassifier 100 trees depth=8 at 1/2/4/8 MI355X",
            "value": value,
            "unit": "rows/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp32",
            "data": "synthetic",
            "config": {
                "model": f"GBMClassifier(bernoulli, {args.trees} trees, depth={args.max_depth}, bins={args.max_bins})",
                "global_batch": global_rows,
                "rows_per_gpu": n,
                "features": f,
                "seq_len": None,
                "parallelism": f"dp{world}",
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
