"""Distributed statistics over row-sharded tensors.

Replaces the Spark SQL aggregates / ``approxQuantile`` actions of the
reference (DummyRegressor.scala:113-129, GBMRegressor.scala:305-309,347-353)
with GPU reductions + RCCL scalar all-reduces.

``dist_quantile`` is an iteratively-refined histogram quantile: global
[min, max] via all-reduce, then ``iters`` rounds of a 2048-bin weighted
histogram all-reduce narrowing the bracket — precision (max-min)/2048^iters,
far below the reference's default approxQuantile tolerance.
"""

from __future__ import annotations

from typing import Optional

import torch

from ..parallel import Comm, get_comm


def dist_mean(values: torch.Tensor, comm: Optional[Comm] = None) -> float:
    comm = comm or get_comm()
    s = float(values.sum())
    n = float(values.numel())
    s = comm.all_reduce_scalar(s)
    n = comm.all_reduce_scalar(n)
    return s / max(n, 1.0)


def dist_weighted_mean(
    values: torch.Tensor, weights: torch.Tensor, comm: Optional[Comm] = None
) -> float:
    comm = comm or get_comm()
    s = comm.all_reduce_scalar(float((values * weights).sum()))
    w = comm.all_reduce_scalar(float(weights.sum()))
    return s / max(w, 1e-300)


def dist_quantile(
    values: torch.Tensor,
    q: float,
    weights: Optional[torch.Tensor] = None,
    comm: Optional[Comm] = None,
    iters: int = 3,
    bins: int = 2048,
) -> float:
    comm = comm or get_comm()
    v = values.float().flatten()
    if weights is None:
        w = torch.ones_like(v)
    else:
        w = weights.float().flatten()
    lo = comm.all_reduce_scalar(float(v.min()) if v.numel() else float("inf"), "min")
    hi = comm.all_reduce_scalar(float(v.max()) if v.numel() else float("-inf"), "max")
    if not (hi > lo):
        return lo
    total = comm.all_reduce_scalar(float(w.sum()))
    target = q * total
    # mass strictly below the current bracket
    below = 0.0
    for _ in range(iters):
        width = (hi - lo) / bins
        if width <= 0:
            break
        idx = ((v - lo) / width).floor().clamp_(0, bins - 1).long()
        inside = (v >= lo) & (v <= hi)
        # f32 index_add_ (f64 global atomics CAS-loop on gfx950), but
        # chunked so no single f32 accumulation exceeds ~2^22 increments
        # (past 2^24 a bin would silently drop unit adds — ADVICE r01),
        # then summed into f64 before the all-reduce.
        idx_in = idx[inside]
        w_in = w[inside].float()
        hist = torch.zeros(bins, dtype=torch.float64, device=v.device)
        chunk = 1 << 22
        for s in range(0, max(idx_in.numel(), 1), chunk):
            part = torch.zeros(bins, dtype=torch.float32, device=v.device)
            part.index_add_(0, idx_in[s:s + chunk], w_in[s:s + chunk])
            hist += part.double()
        comm.all_reduce_(hist)
        cum = below + hist.cumsum(0)
        sel = int((cum >= target).to(torch.int8).argmax())
        below = below + float(hist[:sel].sum())
        lo, hi = lo + sel * width, lo + (sel + 1) * width
    return 0.5 * (lo + hi)
