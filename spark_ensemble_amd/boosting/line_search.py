"""Stage-weight line search for GBM.

Re-creates the reference's distributed line search (GBMLoss.scala:34-76
``GBMLossAggregator`` + ``RDDLossFunction``, driven by commons-math3 Brent
for scalar dim — GBMRegressor.scala:398-425 — and Breeze LBFGS-B with
bounds [0, inf) for vector dim — GBMClassifier.scala:290-292,413-431).

MI355X shape: each function evaluation is one fused pass over the per-GPU
shard computing  sum_i w_i * loss(y_i, p_i + a (.) d_i)  and the per-dim
gradient  sum_i w_i * d_ij * dloss/dp_ij ,  then ONE RCCL all-reduce of the
[1 + dim] payload; the host optimizer (scipy Brent / L-BFGS-B) consumes the
globally-reduced values, so every rank runs the optimizer on identical
numbers and stays in lockstep with no further synchronization.
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..parallel import Comm
from .losses import GBMLoss

# instrumentation: number of fused evaluations the last 1-D search used
# (each is a kernel + all-reduce + host sync — the per-round cost scales
# with it); read by the GBM fit loops for per-round logging
LAST_EVALS = 0


def _eval(loss, label, pred, direction, weight, coeff, want_hess=False):
    """Returns (weighted loss sum, per-dim gradient sums[, hessian sum]) as
    a flat tensor [1 + dim (+1)] on the data's device.  On GPU this is ONE
    fused kernel pass (csrc line_search_eval) instead of ~6 eager passes.
    ``want_hess`` (scalar dim only) additionally accumulates
    sum_i w_i d_i^2 loss''(p_i + a d_i) for the Newton weight search."""
    if pred.is_cuda and pred.shape[1] <= 8:
        from ..ops import dispatch

        m = dispatch._require_hip("line_search_eval")
        if m is not None:
            D = pred.shape[1]
            if isinstance(coeff, torch.Tensor):
                ct = coeff.reshape(-1).float().to(pred.device)
            else:
                ct = torch.full((D,), float(coeff), device=pred.device)
            payload = torch.zeros(
                1 + D + (1 if want_hess else 0),
                dtype=torch.float32, device=pred.device,
            )
            m.line_search_eval(
                payload, label.contiguous(), pred.contiguous(),
                direction.contiguous(), weight.contiguous(), ct,
                loss.loss_id, float(loss.param), bool(want_hess),
            )
            return payload
    p = pred + direction * coeff
    l = (loss.loss(label, p) * weight).sum()
    g = loss.gradient(label, p) * direction * weight.unsqueeze(1)
    out = [l.reshape(1), g.sum(dim=0)]
    if want_hess:
        h = loss.hessian(label, p) * direction * direction * weight.unsqueeze(1)
        out.append(h.sum(dim=0))
    return torch.cat(out)


def optimize_weight_1d(
    loss: GBMLoss,
    label: torch.Tensor,  # [N, 1] encoded
    pred: torch.Tensor,  # [N, 1]
    direction: torch.Tensor,  # [N, 1]
    weight: torch.Tensor,  # [N]
    comm: Optional[Comm] = None,
    max_iter: int = 100,
    tol: float = 1e-6,
    lo: float = 0.0,
    hi: float = 100.0,
) -> float:
    """1-D stage-weight minimization on [0, 100] (the reference's Brent
    SearchInterval — GBMRegressor.scala:413-421).

    For losses with a CONTINUOUS second derivative (squared, logcosh,
    scaledlogcosh, exponential, bernoulli) a safeguarded Newton iteration
    on phi'(a) converges in 3-6 fused evaluations instead of Brent's
    ~25-40 — same minimizer, far fewer host<->device round trips (each
    evaluation is a kernel launch + an RCCL all-reduce + a sync).  Brent
    remains for the non-smooth losses (absolute, huber, quantile) and as
    the fallback when Newton fails to bracket."""
    global LAST_EVALS
    if getattr(loss, "smooth", False) and loss.has_hessian:
        a = _newton_1d(loss, label, pred, direction, weight, comm,
                       max_iter, tol, lo, hi)
        if a is not None:
            return a
    from scipy.optimize import minimize_scalar

    cache = {}

    def phi(a: float) -> float:
        a = float(a)
        if a not in cache:
            payload = _eval(loss, label, pred, direction, weight, a)
            if comm is not None and comm.is_distributed:
                comm.all_reduce_(payload)
            v = float(payload[0])
            if not np.isfinite(v):
                # overflowed exp-style loss: steer Brent back toward the
                # finite region instead of feeding it inf/nan (golden-section
                # otherwise converges to the boundary)
                v = 1e30 * (1.0 + a)
            cache[a] = v
        return cache[a]

    res = minimize_scalar(
        phi,
        bounds=(lo, hi),
        method="bounded",
        options={"maxiter": max_iter, "xatol": max(tol, 1e-8)},
    )
    LAST_EVALS = len(cache)
    return float(res.x)


def optimize_weight_nd(
    loss: GBMLoss,
    label: torch.Tensor,  # [N, D]
    pred: torch.Tensor,  # [N, D]
    direction: torch.Tensor,  # [N, D]
    weight: torch.Tensor,  # [N]
    comm: Optional[Comm] = None,
    max_iter: int = 100,
    tol: float = 1e-6,
    x0: Optional[np.ndarray] = None,
) -> np.ndarray:
    """L-BFGS-B with per-dim bounds [0, inf), started at ``x0`` (default
    ones(dim), the reference's start — GBMClassifier.scala:427).  Callers
    may warm-start from the previous round's solution: the minimizer of
    the convex per-round problem is start-independent, and stage weights
    drift slowly across rounds, so the search converges in fewer
    full-data evaluations (each is a kernel + all-reduce + host sync)."""
    from scipy.optimize import minimize

    dim = pred.shape[1]
    dev = pred.device

    def f(a_np):
        a = torch.from_numpy(a_np.astype(np.float32)).to(dev)
        payload = _eval(loss, label, pred, direction, weight, a)
        if comm is not None and comm.is_distributed:
            comm.all_reduce_(payload)
        v = float(payload[0])
        g = payload[1:].cpu().double().numpy()
        if not np.isfinite(v):
            v = 1e30 * (1.0 + float(np.abs(a_np).sum()))
            g = np.nan_to_num(g, nan=0.0, posinf=1e30, neginf=-1e30)
        return v, g

    start = np.ones(dim) if x0 is None else np.clip(np.asarray(x0, float), 0.0, None)
    res = minimize(
        f,
        start,
        jac=True,
        method="L-BFGS-B",
        bounds=[(0.0, None)] * dim,
        options={"maxiter": max_iter, "ftol": tol, "gtol": tol},
    )
    return res.x.astype(np.float64)


_CHAIN_K = [4]  # adaptive device-chain length (last search's evals)


def _newton_1d(loss, label, pred, direction, weight, comm, max_iter, tol,
               lo, hi):
    """Safeguarded Newton on phi'(a) over [lo, hi]; returns None to fall
    back to Brent (non-finite values, no interior minimum)."""

    def eval_at(a):
        payload = _eval(loss, label, pred, direction, weight, float(a),
                        want_hess=True)
        if comm is not None and comm.is_distributed:
            comm.all_reduce_(payload)
        pc = payload.cpu()  # one sync for all three scalars
        return float(pc[0]), float(pc[1]), float(pc[2])

    global LAST_EVALS
    a = 1.0  # natural stage weight
    blo, bhi = lo, hi
    best_a, best_f = None, float("inf")
    LAST_EVALS = 0
    total = max(8, min(max_iter, 20))

    # GPU fast path: a device-CHAINED sequence of (eval -> safeguarded
    # update) kernels — the update runs on device and feeds the next
    # eval's alpha, so the whole search costs ONE host round trip
    # instead of one per iteration.  Chain length adapts to the previous
    # search's eval count (steady state: zero wasted evals).  Disabled
    # when distributed (the per-eval all-reduce needs the host).
    if (pred.is_cuda and pred.shape[1] == 1
            and (comm is None or not comm.is_distributed)):
        from ..ops import dispatch

        m = dispatch._require_hip("newton_chain_1d")
        if m is not None:
            k_chain = min(8, max(2, _CHAIN_K[0]))
            state = torch.tensor(
                [1.0, lo, hi, 0.0, float("inf"), 0.0, 0.0],
                dtype=torch.float32, device=pred.device,
            )
            payloads = torch.zeros(k_chain, 3, dtype=torch.float32,
                                   device=pred.device)
            m.newton_chain_1d(
                state, payloads, label.contiguous(), pred.contiguous(),
                direction.contiguous(), weight.contiguous(),
                loss.loss_id, float(loss.param), float(max(tol, 1e-8)),
                k_chain,
            )
            st = state.cpu()
            evals, done = int(st[5]), int(st[6])
            LAST_EVALS = evals
            _CHAIN_K[0] = max(2, evals + (0 if done == 1 else 2))
            if done == 1:
                return float(st[0])
            if done == 2:
                return None  # non-finite: Brent fallback
            # chain exhausted before convergence: resume on the host
            a, blo, bhi = float(st[0]), float(st[1]), float(st[2])
            if evals:
                best_f, best_a = float(st[4]), float(st[3])

    for _ in range(total - LAST_EVALS):
        f, g, h = eval_at(a)
        LAST_EVALS += 1
        if not (np.isfinite(f) and np.isfinite(g) and np.isfinite(h)):
            return None
        if f < best_f:
            best_f, best_a = f, a
        # shrink the safeguard bracket using the sign of phi'
        if g > 0:
            bhi = a
        else:
            blo = a
        if abs(g) <= tol * max(1.0, abs(f)) or (bhi - blo) <= tol:
            return a
        if h > 1e-12:
            step = -g / h
            nxt = a + step
            if not (blo < nxt < bhi):
                nxt = 0.5 * (blo + bhi)  # bisect when Newton leaves bracket
            if abs(nxt - a) <= tol * max(1.0, abs(a)):
                return nxt
            a = nxt
        else:
            a = 0.5 * (blo + bhi)
    return best_a
