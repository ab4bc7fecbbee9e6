"""Standalone hist_build microbench (GPU): times the kernel in isolation
across the configurations that matter for the flagship bench, so PMC runs
attribute counters to exactly this kernel."""

import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
from spark_ensemble_amd.ops import dispatch as ops  # noqa: E402


def bench(label, bins, gh, rows, offs, B, iters=3):
    # per-channel maxima precomputed ONCE, as the tree grower does per fit
    # (leaving it to the dispatch default re-runs a strided 10M-row amax
    # per call and used to inflate the C=3 numbers ~2.4x)
    max_abs = gh.abs().amax(dim=0).cpu()
    d = max(1, gh.shape[1] - 2)
    torch.cuda.synchronize()
    # warmup
    out = ops.hist_build(bins, gh, rows, offs, B, d, max_abs)
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(iters):
        out = ops.hist_build(bins, gh, rows, offs, B, d, max_abs)
    torch.cuda.synchronize()
    dt = (time.time() - t0) / iters * 1000
    n_rows = rows.numel()
    gb = n_rows * bins.shape[1] / 1e9
    print(f"{label:42s} {dt:8.2f} ms   ({gb/dt*1000:6.1f} GB/s bin-bytes)")
    return out


def main():
    dev = "cuda:0"
    N, F = 10_000_000, 256
    g = torch.Generator(device=dev).manual_seed(1)
    bins = torch.randint(0, 256, (N, F), generator=g, dtype=torch.uint8, device=dev)
    gh3 = torch.rand(N, 3, generator=g, device=dev)
    rows = torch.arange(N, dtype=torch.int32, device=dev)
    offs1 = torch.tensor([0, N])

    gh2 = gh3[:, :2].contiguous()
    bench("root B=256 C=3 contiguous", bins, gh3, rows, offs1, 256)
    bench("root B=256 C=2 contiguous", bins, gh2, rows, offs1, 256)
    bench("root B=64  C=3 contiguous", (bins & 63), gh3, rows, offs1, 64)

    # shuffled rows (deep-level access pattern)
    perm = torch.randperm(N, generator=g, device=dev).to(torch.int32)
    bench("root B=256 C=2 shuffled", bins, gh2, perm, offs1, 256)
    bench("root B=256 C=3 shuffled", bins, gh3, perm, offs1, 256)

    # 64-node level, half the rows
    n_nodes = 64
    seg = N // 2 // n_nodes
    offs = torch.tensor([i * seg for i in range(n_nodes + 1)])
    rows_half = perm[: seg * n_nodes]
    bench("64 nodes B=256 C=2 shuffled half-rows", bins, gh2, rows_half, offs, 256)
    bench("64 nodes B=256 C=3 shuffled half-rows", bins, gh3, rows_half, offs, 256)


if __name__ == "__main__":
    main()
