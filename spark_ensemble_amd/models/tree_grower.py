"""Histogram-based decision-tree trainer — the compute heart of the framework.

The reference delegates all tree fitting to MLlib's DecisionTree via
``fitBaseLearner`` (reference ensembleParams.scala:64-81); distributed tree
building lives inside Spark there.  This module is the MI355X-native design:

  * features quantile-binned ONCE per fit into a row-major uint8 matrix
    [N, F] (bin id per cell) resident in HBM,
  * level-synchronous growth: one fused histogram pass per level builds
    per-(node, feature, bin) {grad-vector, hessian, count} sums for every
    active node (HIP kernel with LDS-staged bins on gfx950; torch reference
    on CPU),
  * sibling-subtraction: below the root only the SMALLER child of each split
    is histogrammed; the other is parent - built (halves the dominant cost);
    the bottom level needs no histogram at all (leaf values come from the
    parent's split statistics),
  * multi-GPU: rows are sharded across ranks; the per-level histogram tensor
    is all-reduced over RCCL/xGMI (the heavy collective of this framework,
    SURVEY.md section 2.6), after which every rank takes identical split
    decisions with no further communication,
  * split rule: newton/variance gain  score(G, H) = |G|^2 / (H + lambda);
    a multi-output tree on one-hot targets with this rule IS gini splitting
    (sum of per-class Bernoulli variances = gini impurity), so ONE trainer
    serves DecisionTreeClassifier, DecisionTreeRegressor and GBM stages.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

import torch

from ..ops import dispatch as ops
from ..parallel import Comm


@dataclass
class GrowParams:
    max_depth: int = 5
    max_bins: int = 32
    min_instances_per_node: int = 1
    min_info_gain: float = 0.0
    min_child_weight: float = 0.0
    lam: float = 1e-6  # L2 regularization on leaf values


def ensure_binned(frame, x: torch.Tensor, max_bins: int):
    """Bin ``x`` (cached on the frame so repeated fits over one features
    tensor — every GBM round, every bagged learner — bin once).

    Distributed: every rank must use IDENTICAL cut points (split thresholds
    and histograms are only comparable across ranks then), so rank 0's
    shard-sampled edges are broadcast before binning."""
    cached = frame.cache_get("bins", x, max_bins) if frame is not None else None
    if cached is not None:
        return cached
    edges = ops.quantile_bins(x, max_bins)
    if frame is not None and frame.categorical:
        from ..ensemble.binning import apply_categorical_edges_

        apply_categorical_edges_(edges, frame.categorical)
    from ..parallel import get_comm

    comm = get_comm()
    if comm.is_distributed:
        comm.broadcast_(edges, src=0)
    bins = ops.bin_features(x, edges)
    if frame is not None:
        frame.cache_put("bins", x, max_bins, (edges, bins))
    return edges, bins


def grow_tree(
    bins: torch.Tensor,  # [N, F] uint8
    edges: torch.Tensor,  # [F, B-1] f32
    grad: torch.Tensor,  # [N, D] f32 (weighted targets)
    hess: torch.Tensor,  # [N] f32 (weights / newton hessians)
    params: GrowParams,
    comm: Optional[Comm] = None,
    row_mask: Optional[torch.Tensor] = None,
    hess_is_count: Optional[bool] = None,
    train_pred_out: Optional[list] = None,
    gh_max_in: Optional[torch.Tensor] = None,
    root_tot_in: Optional[torch.Tensor] = None,
) -> Dict[str, torch.Tensor]:
    """Grow one tree; returns flat node arrays:

    feature [n] int32 (-1 leaf), threshold [n] f32, left_child [n] int32
    (right = left+1), leaf_value [n, D] f32.

    ``train_pred_out``: pass a list to ALSO receive the training-row
    predictions [N, D] (appended) — training already partitions every row
    to its leaf, so this is a leaf-value scatter instead of the full tree
    walk GBM would otherwise pay per round (only produced when row_mask is
    None, i.e. the tree saw every row).

    Delegates to the vectorized ``grow_forest`` with T=1 whenever the
    channel budget allows (D + hess/count channels <= 8): the arena-based
    grower keeps all per-level bookkeeping tensorized, while this module's
    loop implementation (``_grow_tree_seq``) spends ~5 ms/tree at depth 8
    in per-node Python loops and many-segment ``torch.cat`` copies
    (measured on the flagship bench: ~9.7 ms of GPU idle per 20 ms round,
    concentrated in the deep levels and the leaf-capture finalization).
    The loop version remains the reference for the fused-vs-sequential
    parity tests and the fallback for wide multi-output (gini D > 6)
    trees.
    """
    D = grad.shape[1]
    if hess_is_count is None:
        hess_is_count = bool((hess == 1).all())
    if D + (1 if hess_is_count else 2) <= 8:
        root_rows = None
        if row_mask is not None:
            root_rows = [row_mask.nonzero(as_tuple=True)[0].to(torch.int32)]
        tp: Optional[list] = (
            [] if (train_pred_out is not None and row_mask is None and D == 1)
            else None
        )
        trees = grow_forest(
            bins, edges, grad.unsqueeze(1), hess, params, comm,
            hess_is_count, tp, gh_max_in, root_rows,
            root_tot_in=(
                root_tot_in.reshape(1, -1)
                if root_tot_in is not None and row_mask is None else None
            ),
        )
        if train_pred_out is not None and tp:
            train_pred_out.append(tp[0])  # [N, 1] == [N, D]
        return trees[0]
    return _grow_tree_seq(bins, edges, grad, hess, params, comm, row_mask,
                          hess_is_count, train_pred_out, gh_max_in,
                          root_tot_in)


def _grow_tree_seq(
    bins: torch.Tensor,
    edges: torch.Tensor,
    grad: torch.Tensor,
    hess: torch.Tensor,
    params: GrowParams,
    comm: Optional[Comm] = None,
    row_mask: Optional[torch.Tensor] = None,
    hess_is_count: Optional[bool] = None,
    train_pred_out: Optional[list] = None,
    gh_max_in: Optional[torch.Tensor] = None,
    root_tot_in: Optional[torch.Tensor] = None,
) -> Dict[str, torch.Tensor]:
    """Loop-bookkeeping single-tree grower (see grow_tree docstring)."""
    device = bins.device
    N, F = bins.shape
    D = grad.shape[1]
    B = params.max_bins

    # Count channel is only materialized when hess cannot serve as the row
    # count (non-unit weights); with unit hessians (the default gini /
    # gradient-mode path) C = D + 1 and the hess channel doubles as count —
    # one fewer LDS atomic per cell in the histogram kernel.
    if hess_is_count is None:
        hess_is_count = bool((hess == 1).all())
    if hess_is_count:
        gh = torch.cat([grad, hess.unsqueeze(1)], dim=1).contiguous()
    else:
        cnt = torch.ones(N, dtype=torch.float32, device=device)
        gh = torch.cat(
            [grad, hess.unsqueeze(1), cnt.unsqueeze(1)], dim=1
        ).contiguous()
    C = gh.shape[1]
    idx_c = C - 1
    # per-channel abs maxima for the kernel's fixed-point quantization
    # (callers pass precomputed values from their fused stats pass; the
    # fallback costs one device sync per fit)
    if gh_max_in is not None:
        gh_max = gh_max_in
    else:
        gh_max = gh.abs().amax(dim=0).cpu() if bins.is_cuda else None

    if row_mask is not None:
        row_idx = row_mask.nonzero(as_tuple=True)[0].to(torch.int32)
    else:
        row_idx = torch.arange(N, dtype=torch.int32, device=device)

    # flat node arrays, appended as nodes are allocated
    feats: List[int] = []
    thrs: List[float] = []
    lefts: List[int] = []
    leaves: List[Optional[torch.Tensor]] = []

    def alloc_nodes(k: int) -> int:
        start = len(feats)
        feats.extend([-1] * k)
        thrs.extend([0.0] * k)
        lefts.extend([-1] * k)
        leaves.extend([None] * k)
        return start

    root = alloc_nodes(1)
    # gain-based feature importances (MLlib featureImportances analog):
    # sum of split gains per feature, normalized at the end
    fi = torch.zeros(F, dtype=torch.float64)

    # ----- root totals (one tiny all-reduce; skipped when the caller's
    # fused stats pass already reduced them) ------------------------------
    if root_tot_in is not None:
        root_tot = root_tot_in
    else:
        if row_idx.numel() == N:
            root_tot = gh.sum(dim=0)
        else:
            root_tot = gh.index_select(0, row_idx.long()).sum(dim=0)
        if comm is not None:
            comm.all_reduce_(root_tot)
        root_tot = root_tot.cpu()

    # active level state
    node_ids = [root]
    offsets = torch.tensor([0, row_idx.numel()], dtype=torch.int64)
    totals = root_tot.unsqueeze(0)  # [n_active, C] GLOBAL stats, cpu
    # parent-level histograms (device, GLOBAL sums) for sibling subtraction
    hists: Optional[torch.Tensor] = None
    parent_of: List[int] = []  # active idx -> parent's active idx in prev level
    built_mask: List[bool] = []  # whether this node's hist must be built

    edges_cpu = edges.cpu()

    import os as _os
    import time as _time
    _prof = _os.environ.get("SEA_GROW_PROF") == "1"
    _t = {}

    def _tick(name, t0):
        if _prof:
            _t[name] = _t.get(name, 0.0) + (_time.perf_counter() - t0)

    # leaf-wise (rows, value) capture for train_pred_out
    pred_rows: List[torch.Tensor] = []
    pred_vals: List[torch.Tensor] = []
    capture = train_pred_out is not None and row_mask is None

    def _capture_leaves(act_idx, offs, ridx, ids):
        if not capture:
            return
        ol = offs.tolist() if isinstance(offs, torch.Tensor) else offs
        for i in act_idx:
            s0, e0 = int(ol[i]), int(ol[i + 1])
            if e0 > s0:
                pred_rows.append(ridx[s0:e0])
                pred_vals.append(leaves[ids[i]])

    for depth in range(params.max_depth + 1):
        n_active = len(node_ids)
        if n_active == 0:
            break
        if depth == params.max_depth:
            _finalize_leaves(node_ids, totals, leaves, params, D)
            _capture_leaves(range(n_active), offsets, row_idx, node_ids)
            break

        _t0 = _time.perf_counter() if _prof else 0.0
        split_args = dict(
            lam=params.lam,
            min_child_weight=params.min_child_weight,
            min_instances=params.min_instances_per_node,
            min_info_gain=params.min_info_gain,
            d_dims=D,
        )
        # ----- histograms for this level (+ pipelined reduce + split) ----
        if hists is None:
            # root level: build everything
            new_h = ops.hist_build(
                bins, gh, row_idx, offsets, B, D, gh_max,
                identity_rows=(row_mask is None and bins.is_cuda),
            )
            hists, gain, feat, b, left_stats = _finish_level_split(
                new_h, None, None, None, None, None, n_active, (F, B, C),
                device, comm, split_args,
            )
        else:
            # build only the flagged (smaller) children, then subtract
            built_idx = [j for j in range(n_active) if built_mask[j]]
            bh = None
            if built_idx:
                off_list = offsets.tolist()
                b_off = [0]
                segs = []
                for j in built_idx:
                    s, e = off_list[j], off_list[j + 1]
                    segs.append((s, e))
                    b_off.append(b_off[-1] + (e - s))
                build_rows = (
                    torch.cat([row_idx[s:e] for s, e in segs])
                    if len(segs) > 1
                    else row_idx[segs[0][0] : segs[0][1]]
                )
                bh = ops.hist_build(
                    bins, gh, build_rows, torch.tensor(b_off, dtype=torch.int64),
                    B, D, gh_max,
                )
            nb = [j for j in range(n_active) if not built_mask[j]]
            hists, gain, feat, b, left_stats = _finish_level_split(
                bh,
                torch.as_tensor(built_idx, dtype=torch.long, device=device),
                hists,
                torch.as_tensor(nb, dtype=torch.long, device=device),
                torch.as_tensor([j ^ 1 for j in nb], dtype=torch.long,
                                device=device),
                torch.as_tensor([parent_of[j] for j in nb], dtype=torch.long,
                                device=device),
                n_active, (F, B, C), device, comm, split_args,
            )

        _tick("hist+split", _t0)
        _t0 = _time.perf_counter() if _prof else 0.0
        # overlap: on GPU launch the partition straight off the device
        # split outputs, BEFORE the host fetch below drains the stream —
        # one device sync per level instead of two
        pr_async = None
        if bins.is_cuda:
            pr_async = ops.partition_rows_async(bins, row_idx, offsets, feat, b)

        _tick("part_launch", _t0)
        _t0 = _time.perf_counter() if _prof else 0.0
        gain_cpu = gain.cpu()
        feat_cpu = feat.cpu()
        b_cpu = b.cpu()
        left_stats = left_stats.cpu()

        _tick("sync_fetch", _t0)
        _t0 = _time.perf_counter() if _prof else 0.0
        do_split = torch.isfinite(gain_cpu)
        ns_idx = (~do_split).nonzero(as_tuple=True)[0]
        if ns_idx.numel():
            _finalize_leaves(
                [node_ids[i] for i in ns_idx.tolist()],
                totals[ns_idx],
                leaves,
                params,
                D,
            )
            _capture_leaves(ns_idx.tolist(), offsets, row_idx, node_ids)
        if not bool(do_split.any()):
            break

        # record splits, allocate children
        split_feat = torch.where(do_split, feat_cpu, torch.full_like(feat_cpu, -1))
        child_ids: List[Optional[Tuple[int, int]]] = []
        for i in range(n_active):
            if bool(do_split[i]):
                f = int(feat_cpu[i])
                t = int(b_cpu[i])
                nid = node_ids[i]
                feats[nid] = f
                fi[f] += float(gain_cpu[i])
                thrs[nid] = float(edges_cpu[f, t])
                cid = alloc_nodes(2)
                lefts[nid] = cid
                child_ids.append((cid, cid + 1))
            else:
                child_ids.append(None)

        _tick("leaf+record", _t0)
        _t0 = _time.perf_counter() if _prof else 0.0
        # partition rows of splitting nodes
        if pr_async is not None:
            new_rows, new_offs, _ = ops.partition_rows_finish(*pr_async)
        else:
            new_rows, new_offs, _ = ops.partition_rows(
                bins, row_idx, offsets,
                split_feat.to(torch.int32), b_cpu.to(torch.int32),
            )
        offs_list = new_offs.tolist()

        _tick("part_finish", _t0)
        _t0 = _time.perf_counter() if _prof else 0.0
        # ----- next level bookkeeping ------------------------------------
        next_nodes: List[int] = []
        next_off: List[int] = [0]
        next_tot_rows: List[torch.Tensor] = []
        nb_mask: List[bool] = []
        nb_parent: List[int] = []
        keep_segs: List[Tuple[int, int]] = []
        active_parent = 0  # index into surviving (split) parents for hist ref
        surviving = [i for i in range(n_active) if child_ids[i] is not None]
        hist_keep = torch.tensor(surviving, dtype=torch.long, device=device)
        for rank_i, i in enumerate(surviving):
            lcid, rcid = child_ids[i]
            l_stats = left_stats[i]
            r_stats = totals[i] - l_stats
            l_cnt = float(l_stats[idx_c])
            r_cnt = float(r_stats[idx_c])
            ls, le, re = offs_list[2 * i], offs_list[2 * i + 1], offs_list[2 * i + 2]
            keep_segs.append((ls, re))
            for (cid, s, e, st, built) in (
                (lcid, ls, le, l_stats, l_cnt <= r_cnt),
                (rcid, le, re, r_stats, l_cnt > r_cnt),
            ):
                next_nodes.append(cid)
                next_off.append(next_off[-1] + (e - s))
                next_tot_rows.append(st)
                nb_mask.append(built)
                nb_parent.append(rank_i)

        if len(keep_segs) == 1:
            s, e = keep_segs[0]
            row_idx = new_rows[s:e]
        else:
            row_idx = torch.cat([new_rows[s:e] for s, e in keep_segs])
        offsets = torch.tensor(next_off, dtype=torch.int64)
        node_ids = next_nodes
        totals = torch.stack(next_tot_rows)
        hists = hists.index_select(0, hist_keep)
        parent_of = nb_parent
        built_mask = nb_mask
        _tick("bookkeeping", _t0)

    if _prof:
        print("[grow prof]", {k: round(v * 1000, 2) for k, v in _t.items()})
    # assemble arrays
    n_nodes = len(feats)
    leaf_value = torch.zeros(n_nodes, D, dtype=torch.float32)
    for i, lv in enumerate(leaves):
        if lv is not None:
            leaf_value[i] = lv
    fi_tot = float(fi.sum())
    tree = {
        "feature": torch.tensor(feats, dtype=torch.int32),
        "threshold": torch.tensor(thrs, dtype=torch.float32),
        "left_child": torch.tensor(lefts, dtype=torch.int32),
        "leaf_value": leaf_value,
        "feature_importance": (fi / fi_tot if fi_tot > 0 else fi).to(
            torch.float32
        ),
    }
    if capture and pred_rows:
        rows_cat = torch.cat(pred_rows).long()
        lv_mat = _to_dev_async(
            torch.stack([v.reshape(D) for v in pred_vals]), device
        )
        counts = _to_dev_async(
            torch.tensor([r.numel() for r in pred_rows]), device
        )
        vals = torch.repeat_interleave(lv_mat, counts, dim=0)
        tp = torch.zeros(N, D, dtype=torch.float32, device=device)
        tp[rows_cat] = vals
        train_pred_out.append(tp)
    return {k: _to_dev_async(v, device) for k, v in tree.items()}


MAX_FUSED_TREES = 32  # bounds the [T*N] row-index arena per fused batch


def _finish_level_split(
    built: Optional[torch.Tensor],  # [n_built, F, B, C] LOCAL sums (or None)
    built_t: Optional[torch.Tensor],  # device long: built active indices
    parent_hists: Optional[torch.Tensor],  # [n_parent, F, B, C] GLOBAL
    nb_t: Optional[torch.Tensor],   # device long: non-built active indices
    sib_t: Optional[torch.Tensor],  # device long: their (built) siblings
    par_t: Optional[torch.Tensor],  # device long: their parent rows
    n_active: int,
    shape,                          # (F, B, C)
    device,
    comm: Optional[Comm],
    split_args: dict,
    split_mask: Optional[torch.Tensor] = None,  # [n_active, F] 1/0
):
    """All-reduce the built histograms, assemble the level's full hist
    tensor (scatter + sibling subtraction) and run split_search.

    ``split_mask`` implements per-node FEATURE SUBSPACES without slicing
    the binned matrix: the split-search input is multiplied by the mask,
    so a banned feature's histogram is all-zero and every candidate split
    on it fails the min_instances>=1 child check — it can never win.  The
    UNMASKED histograms are kept for sibling subtraction.

    Distributed: the reduce is FEATURE-CHUNKED and pipelined — chunk c's
    all-reduce overlaps chunk c-1's assembly + split-search compute
    (VERDICT r01 #3a: on 8 GPUs the multi-MB per-level reduce otherwise
    serializes against 1-2 ms of level compute).  Every rank issues the
    identical chunk sequence, so collective ordering is preserved.

    Returns (hists, gain, feat, bin, left_stats); ``hists`` is the GLOBAL
    level histogram (kept for next level's sibling subtraction).
    """
    from ..ops import dispatch as ops_mod

    F, B, C = shape
    root = parent_hists is None

    def assemble_chunk(hists, ch, f0, f1):
        if root:
            hists[:, f0:f1] = ch
            return
        if built_t is not None and built_t.numel():
            hists[built_t, f0:f1] = ch
        if nb_t is not None and nb_t.numel():
            hists[nb_t, f0:f1] = (
                parent_hists[par_t, f0:f1] - hists[sib_t, f0:f1]
            )

    pipelined = (
        comm is not None and comm.is_distributed and F >= 32
        and built is not None and built.numel()
    )
    if not pipelined:
        if built is not None and comm is not None:
            comm.all_reduce_(built)
        if root:
            hists = built
        else:
            # every active row is either built (scattered copy) or
            # non-built (parent - sibling) — assemble_chunk overwrites
            # ALL of it, so no zero fill
            hists = torch.empty(n_active, F, B, C, dtype=torch.float32,
                                device=device)
            if built is not None and built.numel():
                assemble_chunk(hists, built, 0, F)
            else:
                assemble_chunk(hists, built, 0, F)  # subtraction only
        sh = hists
        if split_mask is not None:
            sh = hists * split_mask.unsqueeze(-1).unsqueeze(-1)
        g, ft, b, ls = ops_mod.split_search(sh, **split_args)
        return hists, g, ft, b, ls

    n_chunks = max(2, min(4, F // 16))
    bounds = []
    step = (F + n_chunks - 1) // n_chunks
    for f0 in range(0, F, step):
        bounds.append((f0, min(f0 + step, F)))
    inflight = []
    for f0, f1 in bounds:
        ch = built[:, f0:f1].contiguous()
        inflight.append((f0, f1, ch, comm.all_reduce_async(ch)))
    hists = torch.empty(n_active, F, B, C, dtype=torch.float32, device=device)
    results = []
    for f0, f1, ch, h in inflight:
        h.wait()
        assemble_chunk(hists, ch, f0, f1)
        part = hists[:, f0:f1]
        if split_mask is not None:
            part = part * split_mask[:, f0:f1].unsqueeze(-1).unsqueeze(-1)
        part = part.contiguous()
        g, ft, b, ls = ops_mod.split_search(part, **split_args)
        ft = torch.where(ft >= 0, ft + f0, ft)
        results.append((g, ft, b, ls))
    gains = torch.stack([r[0] for r in results])  # [K, n]
    best = gains.argmax(dim=0)  # first-max tie rule (rank-identical)
    gain = gains.gather(0, best.unsqueeze(0))[0]
    feat = torch.stack([r[1] for r in results]).gather(0, best.unsqueeze(0))[0]
    b_ = torch.stack([r[2] for r in results]).gather(0, best.unsqueeze(0))[0]
    ls_all = torch.stack([r[3] for r in results])  # [K, n, C]
    ls = ls_all.gather(
        0, best.view(1, -1, 1).expand(1, -1, ls_all.shape[2])
    )[0]
    return hists, gain, feat, b_, ls


def grow_forest(
    bins: torch.Tensor,  # [N, F] uint8
    edges: torch.Tensor,  # [F, B-1] f32
    grads: torch.Tensor,  # [N, T] f32 per-tree weighted targets
    hess: torch.Tensor,  # [N] shared or [N, T] per-tree weights
    params: GrowParams,
    comm: Optional[Comm] = None,
    hess_is_count: Optional[bool] = None,
    train_pred_out: Optional[list] = None,
    gh_max_in: Optional[torch.Tensor] = None,
    root_rows: Optional[List[torch.Tensor]] = None,  # per-tree row sets
    feature_masks: Optional[torch.Tensor] = None,  # [T, F] 1/0 subspaces
    root_tot_in: Optional[torch.Tensor] = None,  # [T, C] GLOBAL cpu totals
) -> List[Dict[str, torch.Tensor]]:
    """Grow T single-output trees LEVEL-SYNCHRONOUSLY in fused launches.

    The MI355X replacement for the reference's driver-side parallel fits
    (the K per-class futures of one GBM round, GBMClassifier.scala:377-411,
    and the per-learner futures of Bagging, BaggingRegressor.scala:145-166,
    when every member sees the identity subspace): every active node of
    every tree lands in ONE hist_build launch per level (per-node channel
    selection via gh column offsets), one split_argmax, one partition, and
    — critically for multi-GPU — ONE histogram all-reduce per level
    instead of T.

    grads[:, t] is tree t's weighted regression target (w_t * y_t), hess
    its weight column (shared tensor [N] when all trees reweight rows the
    same way).  Returns T tree dicts (same schema as grow_tree); when
    ``train_pred_out`` is a list, appends the [N, T] training-row
    predictions (leaf scatter).
    """
    device = bins.device
    N, F = bins.shape
    T = grads.shape[1]
    # multi-output trees (gini classification: D = num classes) carry D
    # gradient channels per tree; grads is then [N, T, D]
    D = grads.shape[2] if grads.dim() == 3 else 1
    B = params.max_bins

    # bound the fused batch so the interleaved gh matrix stays modest
    batch_cap = MAX_FUSED_TREES if D == 1 else max(2, (MAX_FUSED_TREES * 3) // (D + 2))
    if T > batch_cap:
        out: List[Dict[str, torch.Tensor]] = []
        preds = [] if train_pred_out is not None else None
        for s in range(0, T, batch_cap):
            sl = slice(s, min(s + batch_cap, T))
            h_sl = hess if hess.dim() == 1 else hess[:, sl].contiguous()
            sub_pred = [] if preds is not None else None
            out.extend(grow_forest(
                bins, edges, grads[:, sl].contiguous(), h_sl, params, comm,
                hess_is_count, sub_pred, gh_max_in,
                root_rows[sl] if root_rows is not None else None,
                feature_masks[sl] if feature_masks is not None else None,
            ))
            if preds is not None:
                preds.append(sub_pred[0])
        if preds is not None:
            train_pred_out.append(torch.cat(preds, dim=1))
        return out

    h_shared = hess.dim() == 1
    if hess_is_count is None:
        hess_is_count = bool((hess == 1).all())
    NN = 1 if hess_is_count else 2
    C = D + NN  # channels per tree
    assert C <= 8, f"fused forest: D + hess/count channels = {C} > 8"
    g3 = grads if grads.dim() == 3 else grads.unsqueeze(2)
    parts = [g3, (hess.unsqueeze(1).expand(N, T) if h_shared else hess)
             .unsqueeze(2)]
    if NN == 2:
        parts.append(torch.ones(N, T, 1, dtype=torch.float32, device=device))
    gh = torch.cat(parts, dim=2).reshape(N, T * C).contiguous()

    if gh_max_in is not None:
        gh_max = gh_max_in  # slot-wise [C] host tensor
    elif bins.is_cuda:
        gm = [float(g3.abs().max())] * D + [float(hess.max())]
        if NN == 2:
            gm.append(1.0)
        gh_max = torch.tensor(gm)
    else:
        gh_max = None

    if root_rows is None:
        assert T * N < 2**31, "fused forest row arena exceeds int32"
        row_idx = torch.arange(N, dtype=torch.int32, device=device).repeat(T)
        root_lens = [N] * T
    else:
        assert len(root_rows) == T
        root_rows = [r.to(device=device, dtype=torch.int32) for r in root_rows]
        row_idx = torch.cat(root_rows) if T > 1 else root_rows[0]
        root_lens = [int(r.numel()) for r in root_rows]
        assert sum(root_lens) < 2**31

    # per-tree node arrays, PREALLOCATED: every per-level decision below
    # is tensorized (fancy-indexed writes over (tree, node) pairs) — the
    # r02 per-node Python loops cost ~80 ms/round at letter shape
    # (26 trees x deep levels) against ~4 ms of kernel work
    max_nodes = 2 ** (params.max_depth + 1) - 1
    feats_t = torch.full((T, max_nodes), -1, dtype=torch.int32)
    thrs_t = torch.zeros(T, max_nodes, dtype=torch.float32)
    lefts_t = torch.full((T, max_nodes), -1, dtype=torch.int32)
    leaves_t = torch.zeros(T, max_nodes, D, dtype=torch.float32)
    cur_len = torch.ones(T, dtype=torch.long)
    fi = torch.zeros(T, F, dtype=torch.float64)

    # root totals: one fused reduction (+ one all-reduce) for all trees
    # (callers with a fused stats pass hand in the already-reduced values,
    # skipping the device sync — same contract as grow_tree's root_tot_in)
    if root_tot_in is not None:
        totals = root_tot_in.cpu().float()
    elif root_rows is None:
        g_sum = g3.sum(dim=0)  # [T, D]
        if h_shared:
            h_sum = hess.sum().reshape(1).expand(T)
        else:
            h_sum = hess.sum(dim=0)
        cnt_col = torch.full((T,), float(N), device=device)
    else:
        gs, hs = [], []
        for t in range(T):
            r = root_rows[t].long()
            gs.append(g3[r, t].sum(dim=0))
            hs.append((hess if h_shared else hess[:, t])[r].sum())
        g_sum = torch.stack(gs)  # [T, D]
        h_sum = torch.stack(hs)
        cnt_col = torch.tensor([float(v) for v in root_lens], device=device)
    if root_tot_in is None:
        cols = [g_sum, h_sum.unsqueeze(1)]
        if NN == 2:
            cols.append(cnt_col.unsqueeze(1))
        root_tot = torch.cat(cols, dim=1)  # [T, C] device
        if comm is not None:
            comm.all_reduce_(root_tot)
        totals = root_tot.cpu()

    # active level state (host tensors; sorted by tree by construction)
    node_tree = torch.arange(T, dtype=torch.long)
    node_nid = torch.zeros(T, dtype=torch.long)
    off_acc = [0]
    for v in root_lens:
        off_acc.append(off_acc[-1] + v)
    offsets = torch.tensor(off_acc, dtype=torch.int64)
    hists: Optional[torch.Tensor] = None
    parent_of_t: Optional[torch.Tensor] = None  # cpu long, per active idx
    built_mask_t: Optional[torch.Tensor] = None  # cpu bool

    edges_cpu = edges.cpu()
    idx_c = C - 1

    capture = train_pred_out is not None and D == 1
    # leaf capture scatters straight into the preallocated [N, T] margin
    # matrix (one fused kernel per finalize — no index lists)
    tp_arena = (
        torch.zeros(N, T, dtype=torch.float32, device=device)
        if capture else None
    )

    # software pipeline: the previous level's arena writes (pure host
    # bookkeeping, read only at final assembly) are DEFERRED until after
    # this level's kernels are queued, so the GPU is never idle behind
    # Python (measured ~110 us of idle per level without this)
    pending_writes = None

    def _finalize_and_capture(idx_t, offs, ridx):
        if idx_t.numel() == 0:
            return
        t_ids = node_tree[idx_t]
        nids = node_nid[idx_t]
        g = totals[idx_t, :D]               # [n, D]
        h = totals[idx_t, D]                # [n]
        denom = (h + params.lam).unsqueeze(1)
        vals = torch.where(denom > 0, g / denom, torch.zeros_like(g))
        leaves_t[t_ids, nids] = vals
        if capture:
            starts = offs[idx_t]
            lens = offs[idx_t + 1] - starts
            nz = (lens > 0).nonzero(as_tuple=True)[0]
            if nz.numel():
                ops.leaf_scatter(tp_arena, ridx, starts[nz], lens[nz],
                                 t_ids[nz], vals[nz, 0])

    for depth in range(params.max_depth + 1):
        n_active = int(node_tree.numel())
        if n_active == 0:
            break
        if depth == params.max_depth:
            _finalize_and_capture(torch.arange(n_active), offsets, row_idx)
            break

        split_args = dict(
            lam=params.lam,
            min_child_weight=params.min_child_weight,
            min_instances=params.min_instances_per_node,
            min_info_gain=params.min_info_gain,
            d_dims=D,
        )
        lvl_mask = None
        if feature_masks is not None:
            lvl_mask = feature_masks.index_select(
                0, _to_dev_async(node_tree, device)
            )
        # ----- fused histograms for this level (+ pipelined reduce) ------
        if hists is None:
            new_h = ops.hist_build_forest(
                bins, gh, row_idx, offsets, (node_tree * C).to(torch.int32),
                B, C, gh_max, d_dims=D,
            )
            hists, gain, feat, b, left_stats = _finish_level_split(
                new_h, None, None, None, None, None, n_active, (F, B, C),
                device, comm, split_args, split_mask=lvl_mask,
            )
        else:
            built_cpu = built_mask_t.nonzero(as_tuple=True)[0]
            nb_cpu = (~built_mask_t).nonzero(as_tuple=True)[0]
            bh = None
            if built_cpu.numel():
                starts = offsets[built_cpu]
                lens = offsets[built_cpu + 1] - starts
                b_off = torch.zeros(built_cpu.numel() + 1, dtype=torch.int64)
                b_off[1:] = torch.cumsum(lens, 0)
                build_rows = ops.gather_ranges(row_idx, starts, lens)
                col0_b = (node_tree[built_cpu] * C).to(torch.int32)
                bh = ops.hist_build_forest(
                    bins, gh, build_rows, b_off, col0_b, B, C, gh_max,
                    d_dims=D,
                )
            hists, gain, feat, b, left_stats = _finish_level_split(
                bh, _to_dev_async(built_cpu, device), hists,
                _to_dev_async(nb_cpu, device),
                _to_dev_async(nb_cpu ^ 1, device),
                _to_dev_async(parent_of_t[nb_cpu], device), n_active,
                (F, B, C), device, comm, split_args, split_mask=lvl_mask,
            )
        # ONE D2H for the level's split results (gain / feat / bin ids
        # are exact in f32 — F < 2^24, B <= 256), queued BEFORE the
        # partition kernel and awaited via an event: the host reads the
        # split outcome the moment split_argmax finishes and does all its
        # bookkeeping WHILE the partition kernel still runs
        pack_d = torch.cat([
            gain, feat.to(torch.float32), b.to(torch.float32),
            left_stats.reshape(-1),
        ])
        ev = None
        if bins.is_cuda:
            pack_h = torch.empty(pack_d.numel(), dtype=torch.float32,
                                 pin_memory=True)
            pack_h.copy_(pack_d, non_blocking=True)
            ev = torch.cuda.Event()
            ev.record()

        pr_async = None
        if bins.is_cuda:
            pr_async = ops.partition_rows_async(bins, row_idx, offsets, feat, b)

        if pending_writes is not None:
            pending_writes()
            pending_writes = None

        if ev is not None:
            ev.synchronize()
            pack = pack_h
        else:
            pack = pack_d
        gain_cpu = pack[:n_active]
        feat_cpu = pack[n_active:2 * n_active].to(torch.int32)
        b_cpu = pack[2 * n_active:3 * n_active].to(torch.int32)
        left_stats = pack[3 * n_active:].reshape(n_active, C)

        do_split = torch.isfinite(gain_cpu)
        ns_idx = (~do_split).nonzero(as_tuple=True)[0]
        _finalize_and_capture(ns_idx, offsets, row_idx)
        if not bool(do_split.any()):
            break

        s_idx = do_split.nonzero(as_tuple=True)[0]  # cpu long, sorted
        t_ids = node_tree[s_idx]
        nids = node_nid[s_idx]
        f_l = feat_cpu[s_idx].long()
        b_l = b_cpu[s_idx].long()
        # child allocation: split nodes are grouped by tree, so the rank
        # within the tree's segment gives each pair's base id (cur_len is
        # read for the NEXT level's ids, so it updates immediately; the
        # arena writes themselves are deferred — see pending_writes)
        counts = torch.bincount(t_ids, minlength=T)
        firsts = torch.cumsum(counts, 0) - counts
        rank = torch.arange(s_idx.numel(), dtype=torch.long) - firsts[t_ids]
        cid = cur_len[t_ids] + 2 * rank
        cur_len += 2 * counts

        def _writes(t_ids=t_ids, nids=nids, f_l=f_l, b_l=b_l,
                    gains=gain_cpu[s_idx].double(), cid=cid):
            feats_t[t_ids, nids] = f_l.to(torch.int32)
            thrs_t[t_ids, nids] = edges_cpu[f_l, b_l]
            fi.index_put_((t_ids, f_l), gains, accumulate=True)
            lefts_t[t_ids, nids] = cid.to(torch.int32)
        pending_writes = _writes

        hists = hists.index_select(0, _to_dev_async(s_idx, device))

        # ----- next level bookkeeping ------------------------------------
        # everything that depends only on the SPLIT results happens here,
        # while the partition kernel is still running; only the offsets
        # math below needs its left counts
        l_st = left_stats[s_idx]                       # [n_split, C]
        r_st = totals[s_idx] - l_st
        totals = torch.stack([l_st, r_st], dim=1).reshape(-1, C)
        l_cnt = l_st[:, idx_c]
        r_cnt = r_st[:, idx_c]
        built_mask_t = torch.stack([l_cnt <= r_cnt, l_cnt > r_cnt],
                                   dim=1).reshape(-1)
        parent_of_t = torch.arange(s_idx.numel(),
                                   dtype=torch.long).repeat_interleave(2)
        node_tree = t_ids.repeat_interleave(2)
        node_nid = torch.stack([cid, cid + 1], dim=1).reshape(-1)

        if pr_async is not None:
            new_rows, new_offs, _ = ops.partition_rows_finish(*pr_async)
        else:
            split_feat = torch.where(do_split, feat_cpu,
                                     torch.full_like(feat_cpu, -1))
            new_rows, new_offs, _ = ops.partition_rows(
                bins, row_idx, offsets,
                split_feat.to(torch.int32), b_cpu.to(torch.int32),
            )
        ls = new_offs[2 * s_idx]
        le = new_offs[2 * s_idx + 1]
        re = new_offs[2 * s_idx + 2]
        lens_pairs = torch.stack([le - ls, re - le], dim=1).reshape(-1)
        offsets = torch.zeros(lens_pairs.numel() + 1, dtype=torch.int64)
        offsets[1:] = torch.cumsum(lens_pairs, 0)
        if int(s_idx.numel()) == n_active:
            row_idx = new_rows
        else:
            row_idx = ops.gather_ranges(new_rows, ls, re - ls)

    if pending_writes is not None:
        pending_writes()

    # assemble per-tree arrays
    trees: List[Dict[str, torch.Tensor]] = []
    fi_tot = fi.sum(dim=1)
    for t in range(T):
        n_nodes = int(cur_len[t])
        tree = {
            "feature": feats_t[t, :n_nodes].clone(),
            "threshold": thrs_t[t, :n_nodes].clone(),
            "left_child": lefts_t[t, :n_nodes].clone(),
            "leaf_value": leaves_t[t, :n_nodes].clone().reshape(n_nodes, D),
            "feature_importance": (
                fi[t] / fi_tot[t] if float(fi_tot[t]) > 0 else fi[t]
            ).to(torch.float32),
        }
        trees.append({k: _to_dev_async(v, device) for k, v in tree.items()})

    if capture:
        train_pred_out.append(tp_arena)
    return trees


def _to_dev_async(t: torch.Tensor, device) -> torch.Tensor:
    """Pinned + non_blocking H2D for small host arrays: a pageable
    .to(device) makes the host WAIT for every queued kernel before the
    copy; async staging keeps the host running ahead."""
    if device is None or torch.device(device).type != "cuda":
        return t.to(device) if device is not None else t
    return t.pin_memory().to(device, non_blocking=True)


def _finalize_leaves(node_ids, totals, leaves, params: GrowParams, D: int):
    t = totals.cpu() if isinstance(totals, torch.Tensor) else totals
    for i, nid in enumerate(node_ids):
        g = t[i, :D]
        h = float(t[i, D])
        leaves[nid] = g / (h + params.lam) if h + params.lam > 0 else g * 0.0
