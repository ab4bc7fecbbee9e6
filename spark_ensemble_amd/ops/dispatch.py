"""Op dispatch: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Policy: on a GPU box the hand-written gfx950 extension (built in-tree from
csrc/ into spark_ensemble_amd/_hip_ops.so) is REQUIRED for the designated
native ops — a missing extension raises instead of silently falling back to
eager torch, so a passing GPU test run always means the HIP path ran.  Set
SEA_ALLOW_EAGER=1 to override for debugging only.
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from . import reference

_hip = None
_hip_error: Optional[str] = None


def _load_hip():
    global _hip, _hip_error
    if _hip is not None or _hip_error is not None:
        return _hip
    try:
        from . import hip_ext

        _hip = hip_ext.load()
    except Exception as e:  # noqa: BLE001
        _hip_error = f"{type(e).__name__}: {e}"
        _hip = None
    return _hip


def hip_available() -> bool:
    return _load_hip() is not None


def _require_hip(op_name: str):
    m = _load_hip()
    if m is None:
        if os.environ.get("SEA_ALLOW_EAGER") == "1":
            return None
        raise RuntimeError(
            f"HIP extension required for {op_name} on GPU but not available "
            f"({_hip_error}). Build it with `python setup.py build_ext --inplace` "
            f"or set SEA_ALLOW_EAGER=1 to run the slow eager fallback."
        )
    return m


# ---------------------------------------------------------------------------


def sample_weights(replacement, ratio, n, seed, device=None, base_weight=None, rank=0):
    dev = torch.device(device) if device is not None else torch.device("cpu")
    if dev.type == "cuda":
        m = _require_hip("sample_weights")
        if m is not None:
            w = torch.empty(n, dtype=torch.float32, device=dev)
            m.sample_weights(w, bool(replacement), float(ratio), int(seed), int(rank))
            if base_weight is not None:
                w *= base_weight
            return w
    return reference.sample_weights(replacement, ratio, n, seed, device, base_weight, rank)


def quantile_bins(x, max_bins=256, sample_rows=262_144, seed=17):
    # sort-based; torch sort is fine on both devices (one-time cost per fit)
    return reference.quantile_bins(x, max_bins, sample_rows, seed)


def bin_features(x, edges):
    if x.is_cuda:
        m = _require_hip("bin_features")
        if m is not None:
            out = torch.empty(x.shape, dtype=torch.uint8, device=x.device)
            m.bin_features(out, x.contiguous(), edges.contiguous())
            return out
    return reference.bin_features(x, edges)


def hist_build(bins, gh, row_idx, node_offsets, num_bins, d_dims=-1, max_abs=None, identity_rows=False):
    """Per-(node, feature, bin) channel sums.

    Channel contract (tree_grower.py): gh[:, :d_dims] are SIGNED gradient
    channels, gh[:, d_dims:] are NON-NEGATIVE hess/count channels — the
    gfx950 kernel accumulates in packed fixed-point u64 LDS cells
    (ds_add_u64 is ~13x faster than ds_add_f32 on gfx950, see
    profiles/r01_hist_probe.md).  ``max_abs`` (host [C] tensor of per-channel
    abs maxima) sizes the quantization; pass it from the caller to avoid a
    device sync per level — when None it is computed here (one sync).
    """
    if bins.is_cuda:
        m = _require_hip("hist_build")
        if m is not None:
            n_nodes = node_offsets.numel() - 1
            F = bins.shape[1]
            C = gh.shape[1]
            if max_abs is None:
                max_abs = gh.abs().amax(dim=0).cpu()
            if C > 8:
                # wide one-hot targets (K-class gini trees, K > 7): the
                # kernel caps at 8 channels, so re-layout gh ONCE into
                # contiguous per-chunk groups [g_s..g_{s+w-1}, tail...]
                # (tail = hess/count duplicated per chunk, <= 2 cols) and
                # launch one kernel per chunk with a column offset — no
                # per-chunk gh copies (r01 chunking index_select'ed the
                # whole gh every chunk).  Chunk width is 8 - len(tail) so
                # every launch is within the channel cap.
                D = d_dims if d_dims > 0 else C - 1
                tail = list(range(D, C))
                step = 8 - len(tail)
                assert 1 <= step <= 7, f"hist_build tail too wide: {len(tail)}"
                spans = []  # (col0_in_wide, w)
                cols: list = []
                for s in range(0, D, step):
                    w = min(step, D - s)
                    spans.append((len(cols), w))
                    cols.extend(range(s, s + w))
                    cols.extend(tail)
                idx = torch.tensor(cols, device=gh.device)
                gh_wide = gh.index_select(1, idx).contiguous()  # once
                ma_wide = max_abs[torch.tensor(cols)]
                parts = []
                for ci, (c0, w) in enumerate(spans):
                    cc = w + len(tail)
                    part = hist_build_forest(
                        bins, gh_wide, row_idx, node_offsets,
                        torch.full((n_nodes,), c0, dtype=torch.int32),
                        num_bins, cc, ma_wide[c0:c0 + cc], d_dims=w,
                    )
                    parts.append(part[..., :w])
                    if ci == len(spans) - 1:
                        parts.append(part[..., w:])  # tail once
                return torch.cat(parts, dim=-1)
            # every (node, f, b) cell is fully written by either the
            # single-chunk flush or the multi-chunk decode — no zero fill
            out = torch.empty(
                n_nodes, F, num_bins, C, dtype=torch.float32, device=bins.device
            )
            m.hist_build(
                out,
                bins,
                gh.contiguous(),
                row_idx.to(torch.int32),
                node_offsets.to(torch.int64).cpu(),
                int(num_bins),
                int(d_dims),
                max_abs.to(torch.float32),
                bool(identity_rows),
                _EMPTY_I32,
                0,
            )
            return out
    return reference.hist_build(bins, gh, row_idx, node_offsets, num_bins)


_EMPTY_I32 = torch.empty(0, dtype=torch.int32)
_EMPTY_U8 = torch.empty(0, dtype=torch.uint8)


def hist_build_forest(bins, gh, row_idx, node_offsets, node_col0, num_bins,
                      c_per_node, max_abs, d_dims=1):
    """Forest histogram build: gh is [N, CH] with contiguous per-node
    channel groups; node_col0[n] selects the base column node n
    accumulates (tree fusion: owning_tree * C; wide-gini chunking: the
    chunk's column offset).  One launch covers every active node of every
    tree in the level (the MI355X form of the reference's parallel
    per-class futures, GBMClassifier.scala:377-411).  ``max_abs`` is
    slot-wise ([c_per_node]): max over the corresponding column of every
    group."""
    if bins.is_cuda:
        m = _require_hip("hist_build")
        if m is not None:
            n_nodes = node_offsets.numel() - 1
            F = bins.shape[1]
            out = torch.empty(
                n_nodes, F, num_bins, c_per_node,
                dtype=torch.float32, device=bins.device,
            )
            m.hist_build(
                out, bins, gh.contiguous(), row_idx.to(torch.int32),
                node_offsets.to(torch.int64).cpu(), int(num_bins),
                int(d_dims),
                max_abs.to(torch.float32),
                False,
                node_col0.to(torch.int32).cpu().contiguous(),
                int(c_per_node),
            )
            return out
    return reference.hist_build_forest(
        bins, gh, row_idx, node_offsets, node_col0, num_bins, c_per_node
    )


def split_search(hist, lam=1e-6, min_child_weight=0.0, min_instances=1.0, min_info_gain=0.0, d_dims=-1):
    if hist.is_cuda:
        c_chk = hist.shape[3]
        if c_chk > 8:
            # wide one-hot channels (K > 7 gini trees): eager torch path
            # (runs on GPU; the fused kernel caps at 8 channels)
            return reference.split_search(
                hist, lam, min_child_weight, min_instances, min_info_gain,
                d_dims,
            )
        m = _require_hip("split_argmax")
        if m is not None:
            n, f, b, c = hist.shape
            gain = torch.empty(n, dtype=torch.float32, device=hist.device)
            feat = torch.empty(n, dtype=torch.int32, device=hist.device)
            bin_ = torch.empty(n, dtype=torch.int32, device=hist.device)
            left_stats = torch.empty(n, c, dtype=torch.float32, device=hist.device)
            m.split_argmax(
                gain, feat, bin_, left_stats, hist.contiguous(),
                int(d_dims), float(lam), float(min_child_weight),
                float(min_instances), float(min_info_gain),
            )
            return gain, feat.long(), bin_.long(), left_stats
    return reference.split_search(hist, lam, min_child_weight, min_instances, min_info_gain, d_dims)


def partition_rows(bins, row_idx, node_offsets, feat, thr):
    if bins.is_cuda:
        m = _require_hip("partition_rows")
        if m is not None:
            new_rows, lc_dev, offs_cpu = partition_rows_async(
                bins, row_idx, node_offsets, feat, thr
            )
            return partition_rows_finish(new_rows, lc_dev, offs_cpu)
    return reference.partition_rows(bins, row_idx, node_offsets, feat, thr)


_BT_CACHE: list = []  # [(weakref(bins), bins_t)] — at most 2 entries


def _bins_transposed(bins):
    """[F, N] transpose of the binned matrix, cached by OBJECT identity
    (the grower reuses one bins tensor across all rounds of a fit).  The
    partition kernel reads it instead of the row-major matrix: a node's
    rows are locally dense, so a 64-B line yields many useful bytes
    instead of one.  One transpose_u8 pass (~0.7 ms at 10M x 256)
    amortizes over every level of every round."""
    import weakref

    m = _load_hip()
    if m is None or not bins.is_cuda:
        return None
    # prune entries whose bins tensor died FIRST: a stale entry pins a
    # multi-GB transpose (25.6 GB at the 100M x 256 capacity point)
    _BT_CACHE[:] = [(w, b) for (w, b) in _BT_CACHE if w() is not None]
    for wr, bt in _BT_CACHE:
        if wr() is bins:
            return bt
    n, f = bins.shape
    bt = torch.empty(f, n, dtype=torch.uint8, device=bins.device)
    m.transpose_u8(bt, bins)
    _BT_CACHE.insert(0, (weakref.ref(bins), bt))
    del _BT_CACHE[1:]
    return bt


def partition_rows_async(bins, row_idx, node_offsets, feat, thr):
    """Launch the partition kernel WITHOUT syncing (feat/thr may be live
    device tensors straight from split_argmax); call
    :func:`partition_rows_finish` after the caller's own device sync."""
    import time as _time

    _prof = os.environ.get("SEA_GROW_PROF") == "1"
    _t0 = _time.perf_counter() if _prof else 0.0
    m = _require_hip("partition_rows")
    n = node_offsets.numel() - 1
    offs_cpu = node_offsets.to(torch.int64).cpu()
    row_idx = row_idx.to(torch.int32)
    new_rows = torch.empty_like(row_idx)
    left_counts = torch.zeros(n, dtype=torch.int32, device=bins.device)
    f32 = feat.to(torch.int32).to(bins.device)
    t32 = thr.to(torch.int32).to(bins.device)
    if _prof:
        _t1 = _time.perf_counter()
    bt = _bins_transposed(bins)
    m.partition_rows(
        new_rows,
        left_counts,
        bins,
        bt if bt is not None else _EMPTY_U8,
        row_idx,
        offs_cpu,
        f32,
        t32,
    )
    # prefetch the left counts: non-blocking D2H + event, so finish()
    # wakes the host the moment the kernel ends (a blocking .cpu() there
    # costs an extra stream round trip after the partition)
    lc_h = torch.empty(left_counts.numel(), dtype=torch.int32,
                       pin_memory=True)
    lc_h.copy_(left_counts, non_blocking=True)
    ev = torch.cuda.Event()
    ev.record()
    if _prof:
        _t2 = _time.perf_counter()
        print(f"[part prof] prep={(_t1-_t0)*1000:.2f} native={(_t2-_t1)*1000:.2f}")
    return new_rows, (lc_h, ev), offs_cpu


def gather_ranges(src, starts, lens):
    """Concatenation of ``src[starts_i : starts_i + lens_i)`` ranges.

    src is an int32 device row arena; starts/lens are SMALL cpu int64
    tensors (one entry per active node).  GPU: one kernel pass (binary
    search over the L2-resident prefix table) instead of torch's
    repeat_interleave + cumsum + arange + gather chain.  CPU: slice cat.
    """
    if src.is_cuda:
        m = _require_hip("gather_ranges")
        if m is not None:
            out = torch.empty(int(lens.sum()), dtype=torch.int32,
                              device=src.device)
            m.gather_ranges(out, src, starts.to(torch.int64),
                            lens.to(torch.int64))
            return out
    st, ln = starts.tolist(), lens.tolist()
    segs = [src[s:s + l] for s, l in zip(st, ln) if l > 0]
    if not segs:
        return torch.empty(0, dtype=src.dtype, device=src.device)
    return segs[0].clone() if len(segs) == 1 else torch.cat(segs)


def leaf_scatter(tp, row_idx, starts, lens, tree, val):
    """tp[row_idx[starts_i : starts_i+lens_i], tree_i] = val_i for every
    segment i — the train-pred leaf capture, fused into one kernel on GPU
    (no device-side index lists are materialized at all)."""
    if tp.is_cuda:
        m = _require_hip("leaf_scatter")
        if m is not None:
            m.leaf_scatter(tp, row_idx, starts.to(torch.int64),
                           lens.to(torch.int64), tree.to(torch.int64),
                           val.to(torch.float32))
            return
    st, ln = starts.tolist(), lens.tolist()
    tl, vl = tree.tolist(), val.tolist()
    for s, l, t, v in zip(st, ln, tl, vl):
        if l > 0:
            tp[row_idx[s:s + l].long(), int(t)] = float(v)


def partition_rows_finish(new_rows, left_counts, offs_cpu):
    n = offs_cpu.numel() - 1
    if isinstance(left_counts, tuple):
        lc_h, ev = left_counts
        ev.synchronize()
        lc = lc_h.to(torch.int64)
    else:
        lc = left_counts.cpu().to(torch.int64)
    sizes = torch.empty(2 * n, dtype=torch.int64)
    seg = offs_cpu[1:] - offs_cpu[:-1]
    sizes[0::2] = lc
    sizes[1::2] = seg - lc
    new_offs = torch.cat([torch.zeros(1, dtype=torch.int64), sizes.cumsum(0)])
    return new_rows, new_offs, lc


def tree_predict(x, feature, threshold, left_child, leaf_value, max_depth):
    if x.is_cuda:
        m = _require_hip("tree_predict")
        if m is not None:
            N = x.shape[0]
            D = leaf_value.shape[1]
            out = torch.empty(N, D, dtype=torch.float32, device=x.device)
            m.tree_predict(
                out,
                x.contiguous(),
                feature.to(torch.int32).to(x.device),
                threshold.to(torch.float32).to(x.device),
                left_child.to(torch.int32).to(x.device),
                leaf_value.to(torch.float32).to(x.device).contiguous(),
            )
            return out
    return reference.tree_predict(x, feature, threshold, left_child, leaf_value, max_depth)


def forest_predict(x, trees, weights=None, max_depth=64, cache=None):
    """Packed-forest ensemble inference.

    ``cache``: an optional caller-owned dict; the packed node arena is
    stored there under key "pack" so repeated transform calls on the same
    (immutable, post-fit) tree list skip the per-call packing.
    """
    if x.is_cuda and trees:
        m = _require_hip("forest_predict")
        if m is not None:
            pack = cache.get("pack") if cache is not None else None
            if pack is None or pack["T"] != len(trees) or pack["dev"] != x.device:
                pack = _pack_forest(trees, weights, x.device)
                if cache is not None:
                    cache["pack"] = pack
            return _forest_predict_packed(m, x, pack)
    return reference.forest_predict(x, trees, weights, max_depth)


_FP2_GROUP_NODES = 20352  # ~159 KiB of packed 8-B nodes per LDS group (160 KiB LDS/CU)


def _pack_forest(trees, weights, dev):
    """Pack a tree list into the arena tensors both kernel paths consume.
    Done once per (model, device) — callers cache the result."""
    # per-tree moves BEFORE the cat: a resumed ensemble mixes CPU-loaded
    # checkpoint stages with device-fitted ones
    feats = torch.cat([t["feature"].to(dev, torch.int32) for t in trees])
    thrs = torch.cat([t["threshold"].to(dev, torch.float32) for t in trees])
    lefts = torch.cat([t["left_child"].to(dev, torch.int32) for t in trees])
    leaves = torch.cat(
        [t["leaf_value"].to(dev, torch.float32) for t in trees]
    ).contiguous()
    sizes = [t["feature"].numel() for t in trees]
    sizes_t = torch.tensor(sizes, dtype=torch.int64)
    offsets = torch.cat([torch.zeros(1, dtype=torch.int64), sizes_t.cumsum(0)])[:-1]
    offsets32 = offsets.to(torch.int32).to(dev)
    D = trees[0]["leaf_value"].shape[1]
    T = len(trees)
    if weights is None:
        w = torch.ones(T, dtype=torch.float32, device=dev)
    else:
        w = weights.to(torch.float32).to(dev)

    pack = {
        "T": T, "D": D, "dev": dev, "feats": feats, "thrs": thrs,
        "lefts": lefts, "leaves": leaves, "offsets32": offsets32, "w": w,
        "v2": False,
    }
    fits_v2 = D <= 8 and max(sizes) <= _FP2_GROUP_NODES
    if not fits_v2:
        return pack

    # ---- v2: pack (feat s16 | left s16 | payload f32-bits) into one u64
    # per node; for D == 1 a leaf's payload IS its weighted leaf value so
    # the kernel never touches the leaf tensor (csrc forest_predict2)
    if D == 1:
        tree_id = torch.repeat_interleave(
            torch.arange(T, device=dev), sizes_t.to(dev)
        )
        payload = torch.where(feats < 0, w[tree_id] * leaves[:, 0], thrs)
    else:
        payload = thrs
    node64 = (
        (feats.to(torch.int64) & 0xFFFF)
        | ((lefts.to(torch.int64) & 0xFFFF) << 16)
        | ((payload.view(torch.int32).to(torch.int64) & 0xFFFFFFFF) << 32)
    ).contiguous()

    # greedy contiguous tree groups under the LDS node budget
    groups = []
    off_list = offsets.tolist()
    first = 0
    acc = 0
    max_nodes = 0
    for t in range(T):
        if acc + sizes[t] > _FP2_GROUP_NODES and acc > 0:
            groups.append([first, t - first, off_list[first], acc])
            max_nodes = max(max_nodes, acc)
            first, acc = t, 0
        acc += sizes[t]
    groups.append([first, T - first, off_list[first], acc])
    max_nodes = max(max_nodes, acc)
    pack.update(
        v2=True,
        node64=node64,
        groups_t=torch.tensor(groups, dtype=torch.int32, device=dev),
        max_nodes=max_nodes,
    )
    _pack_forest_binned(pack, feats, thrs, payload)
    return pack


def _pack_forest_binned(pack, feats, thrs, payload):
    """Rank-transform serving structures (EXACT): collect each feature's
    sorted unique threshold set; a node's threshold becomes its RANK in
    that set, and at predict time every row is rank-transformed once via
    the bin_features kernel (lower_bound against the padded set), after
    which  x <= thr  <=>  rank(x) <= rank(thr)  on u8 bins — rows shrink
    4x for the divergent walk gathers.  Disabled when any feature carries
    > 255 distinct thresholds (u8 overflow)."""
    dev = pack["dev"]
    mask = feats >= 0
    m_cnt = int(mask.sum())
    if m_cnt == 0:
        return
    f_int = feats[mask].long()
    t_val = thrs[mask]
    f_max = int(f_int.max()) + 1
    # sort by (feature, threshold)
    order = torch.argsort(t_val)
    f_s, t_s = f_int[order], t_val[order]
    order2 = torch.argsort(f_s, stable=True)
    p = order[order2]
    f_s, t_s = f_s[order2], t_s[order2]
    keep = torch.ones_like(f_s, dtype=torch.bool)
    keep[1:] = (f_s[1:] != f_s[:-1]) | (t_s[1:] != t_s[:-1])
    uidx = keep.long().cumsum(0) - 1
    fu, tu = f_s[keep], t_s[keep]
    counts = torch.bincount(fu, minlength=f_max)
    maxcnt = int(counts.max())
    if maxcnt > 255:
        return
    offs = torch.zeros(f_max + 1, dtype=torch.long, device=dev)
    offs[1:] = counts.cumsum(0)
    rank_sorted = uidx - offs[f_s]
    rank_node = torch.empty(m_cnt, dtype=torch.long, device=dev)
    rank_node[p] = rank_sorted
    rank_full = torch.zeros(feats.numel(), dtype=torch.int64, device=dev)
    rank_full[mask] = rank_node
    payload_bits = torch.where(
        mask,
        rank_full,
        payload.view(torch.int32).to(torch.int64) & 0xFFFFFFFF,
    )
    node64b = (
        (feats.to(torch.int64) & 0xFFFF)
        | ((pack["lefts"].to(torch.int64) & 0xFFFF) << 16)
        | (payload_bits << 32)
    ).contiguous()
    pack.update(
        binned_ok=True, node64b=node64b, bin_fu=fu, bin_tu=tu,
        bin_offs=offs, bin_maxcnt=max(maxcnt, 1), bin_fmax=f_max,
    )


def _binned_edges(pack, F):
    """[F, maxcnt] padded per-feature threshold sets for bin_features
    (pad = +3e38 so lower_bound is unaffected); cached per F."""
    cached = pack.get("bedges")
    if cached is not None and cached.shape[0] == F:
        return cached
    dev = pack["dev"]
    mc = pack["bin_maxcnt"]
    edges = torch.full((F, mc), 3.0e38, dtype=torch.float32, device=dev)
    fu, tu, offs = pack["bin_fu"], pack["bin_tu"], pack["bin_offs"]
    col = torch.arange(fu.numel(), device=dev) - offs[fu]
    edges[fu, col] = tu
    pack["bedges"] = edges
    return edges


def _run_fp2(m, x_in, pack, mode):
    """mode: 'raw' (f32 rows), 'binned' (u8 rank rows), 'binned_t'
    (transposed u8 — wave-coalesced top-level gathers)."""
    out = torch.zeros(x_in.shape[0], pack["D"], dtype=torch.float32,
                      device=x_in.device)
    if mode == "raw":
        m.forest_predict2(out, x_in, pack["node64"], pack["leaves"],
                          pack["offsets32"], pack["w"], pack["groups_t"],
                          pack["D"], pack["max_nodes"], 0)
        return out
    edges = _binned_edges(pack, x_in.shape[1])
    xb = torch.empty(x_in.shape, dtype=torch.uint8, device=x_in.device)
    m.bin_features(xb, x_in, edges)
    if mode == "binned_t":
        xbt = torch.empty(xb.shape[1], xb.shape[0], dtype=torch.uint8,
                          device=xb.device)
        m.transpose_u8(xbt, xb)
        xb = xbt
        m.forest_predict2(out, xb, pack["node64b"], pack["leaves"],
                          pack["offsets32"], pack["w"], pack["groups_t"],
                          pack["D"], pack["max_nodes"], 1)
    else:
        m.forest_predict2(out, xb, pack["node64b"], pack["leaves"],
                          pack["offsets32"], pack["w"], pack["groups_t"],
                          pack["D"], pack["max_nodes"], 0)
    return out


_SERVE_MODES = ("binned_t", "binned", "raw")


def _forest_predict_packed(m, x, pack):
    D = pack["D"]
    if pack["v2"] and x.shape[1] < 32768 and hasattr(m, "forest_predict2"):
        xc = x.contiguous()
        env_mode = os.environ.get("SEA_SERVE_MODE")
        if os.environ.get("SEA_SERVE_RAW") == "1":
            env_mode = "raw"
        binned_avail = (
            pack.get("binned_ok") and pack["bin_fmax"] <= x.shape[1]
        )
        if not binned_avail:
            return _run_fp2(m, xc, pack, "raw")
        if env_mode in _SERVE_MODES:
            return _run_fp2(m, xc, pack, env_mode)
        # which walk mode wins depends on the forest's x-access pattern
        # (measured r02: binned 1.8x faster for a 100-tree GBM, raw 1.7x
        # faster for 50 subspace-bagged trees) — all modes are EXACT, so
        # on the first large batch time each once and remember the winner
        mode = pack.get("mode")
        if mode is None:
            if x.shape[0] < (1 << 20):
                return _run_fp2(m, xc, pack, "binned")  # small: either way
            import time

            probe = xc[: 1 << 19]
            times = {}
            for cand in _SERVE_MODES:
                _run_fp2(m, probe, pack, cand)  # warm
                torch.cuda.synchronize()
                t0 = time.perf_counter()
                _run_fp2(m, probe, pack, cand)
                torch.cuda.synchronize()
                times[cand] = time.perf_counter() - t0
            mode = min(times, key=times.get)
            pack["mode"] = mode
        return _run_fp2(m, xc, pack, mode)
    out = torch.zeros(x.shape[0], D, dtype=torch.float32, device=x.device)
    m.forest_predict(out, x.contiguous(), pack["feats"], pack["thrs"],
                     pack["lefts"], pack["leaves"], pack["offsets32"],
                     pack["w"], D)
    return out


def logreg_loss_grad(x, y_int, w, wmat, has_bias):
    """Single-pass fused logistic loss+gradient.

    Returns payload [1 + (F+1)*K]: [loss_sum, grad(F,K) flat, grad_bias(K)]
    (unnormalized sums; caller all-reduces and divides by total weight).
    On GPU uses the hand-written gfx950 kernel (csrc/linear.hip) when the
    (F, K) shape fits the register budget, else the torch fallback.
    """
    f = x.shape[1]
    k = wmat.shape[1]
    if x.is_cuda:
        m = _require_hip("logreg_loss_grad")
        if m is not None and m.logreg_fused_supported(f, k):
            payload = torch.zeros(
                1 + (f + 1) * k, dtype=torch.float32, device=x.device
            )
            m.logreg_loss_grad(
                payload, x.contiguous(), y_int.contiguous(),
                w.contiguous(), wmat.contiguous(), bool(has_bias),
            )
            return payload
    return reference.logreg_loss_grad(x, y_int, w, wmat, has_bias)
