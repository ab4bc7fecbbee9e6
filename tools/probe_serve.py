"""Serving-kernel probe: forest_predict v1 (global pointer-chase) vs v2
(LDS-staged tree groups, 4-way tree interleave) on synthetic valid
complete depth-8 trees — isolates kernel time from model fitting so a
rocprofv3 --stats run of this script is pure serving evidence.

Usage: python tools/probe_serve.py [rows] [trees] [depth]
"""
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from spark_ensemble_amd.ops import dispatch

DEV = "cuda:0"


def rand_tree(depth, F, D, gen):
    n_internal = 2 ** depth - 1
    n_nodes = 2 ** (depth + 1) - 1
    feature = torch.full((n_nodes,), -1, dtype=torch.int32)
    feature[:n_internal] = torch.randint(0, F, (n_internal,), generator=gen)
    threshold = torch.zeros(n_nodes)
    # quantile-ish thresholds so walks split realistically on randn data
    threshold[:n_internal] = torch.randn(n_internal, generator=gen) * 0.5
    left = torch.full((n_nodes,), -1, dtype=torch.int32)
    left[:n_internal] = 2 * torch.arange(n_internal, dtype=torch.int32) + 1
    leaf = torch.randn(n_nodes, D, generator=gen) * 0.1
    return {"feature": feature.to(DEV), "threshold": threshold.to(DEV),
            "left_child": left.to(DEV), "leaf_value": leaf.to(DEV)}


def timeit(fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / reps


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 10_000_000
    T = int(sys.argv[2]) if len(sys.argv) > 2 else 100
    depth = int(sys.argv[3]) if len(sys.argv) > 3 else 8
    F = 256
    g = torch.Generator().manual_seed(3)
    x = torch.randn(n, F, generator=g).to(DEV)
    trees = [rand_tree(depth, F, 1, g) for _ in range(T)]
    w = torch.ones(T)

    m = dispatch._load_hip()
    assert m is not None

    # v2 packed + cached (binned rank-transform mode when available)
    cache = {}
    t_first = timeit(lambda: dispatch.forest_predict(x, trees, w, cache=cache), reps=1)
    t2 = timeit(lambda: dispatch.forest_predict(x, trees, w, cache=cache))
    out2 = dispatch.forest_predict(x, trees, w, cache=cache)

    # v2 raw-float mode
    import os
    os.environ["SEA_SERVE_RAW"] = "1"
    t2raw = timeit(lambda: dispatch.forest_predict(x, trees, w, cache=cache))
    out2raw = dispatch.forest_predict(x, trees, w, cache=cache)
    del os.environ["SEA_SERVE_RAW"]
    err_raw = float((out2raw - out2).abs().max())

    # v1 via the packed arena fallback entry
    pack = cache["pack"]
    out1 = torch.zeros(n, 1, dtype=torch.float32, device=DEV)

    def v1():
        out1.zero_()
        m.forest_predict(out1, x, pack["feats"], pack["thrs"], pack["lefts"],
                         pack["leaves"], pack["offsets32"], pack["w"], 1)
    t1 = timeit(v1)
    v1()
    err = float((out1 - out2).abs().max())

    print(json.dumps({
        "probe": "serve", "rows": n, "trees": T, "depth": depth,
        "v2_binned_ms": round(t2 * 1000, 2),
        "v2_first_call_ms": round(t_first * 1000, 2),
        "v2_raw_ms": round(t2raw * 1000, 2),
        "v1_ms": round(t1 * 1000, 2),
        "v2_binned_rows_per_sec": round(n / t2),
        "v2_raw_rows_per_sec": round(n / t2raw),
        "v1_rows_per_sec": round(n / t1),
        "binned_vs_raw_max_diff": err_raw,
        "v1_vs_v2_max_diff": err,
    }))


if __name__ == "__main__":
    main()
