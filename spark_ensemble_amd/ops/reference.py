"""Pure-torch reference implementations of every compute op.

These define the semantics that the hand-written CDNA4 HIP kernels
(csrc/*.hip) must reproduce; GPU numerics tests compare kernel output
against these at fp32.  On CPU (no GPU available) they ARE the execution
path, which keeps the whole framework runnable for the non-gpu test suite.

Each op cites the reference behavior it implements (SURVEY.md section 2.7
maps ops -> reference call sites).
"""

from __future__ import annotations

from typing import List, Optional, Tuple

import torch


# ---------------------------------------------------------------------------
# Sampling (reference RDD.sample sites: BaggingRegressor.scala:149-150,
# GBMRegressor.scala:357-359, GBMClassifier.scala:329-331)
# ---------------------------------------------------------------------------


def _gen_for(device, seed: int):
    dev = torch.device(device) if device is not None else torch.device("cpu")
    g = torch.Generator(device=dev)
    g.manual_seed(int(seed) & 0x7FFFFFFFFFFFFFFF)
    return g, dev


def sample_weights(
    replacement: bool,
    ratio: float,
    n: int,
    seed: int,
    device=None,
    base_weight: Optional[torch.Tensor] = None,
    rank: int = 0,
) -> torch.Tensor:
    """Poisson(ratio) multiplicities (with replacement) or Bernoulli(ratio)
    0/1 mask (without), as an instance-weight vector."""
    g, dev = _gen_for(device, seed * 1_000_003 + rank)
    if ratio >= 1.0 and not replacement:
        w = torch.ones(n, dtype=torch.float32, device=dev)
    elif replacement:
        rate = torch.full((n,), float(ratio), device=dev)
        w = torch.poisson(rate, generator=g)
    else:
        w = (torch.rand(n, generator=g, device=dev) < ratio).float()
    if base_weight is not None:
        w = w * base_weight
    return w


# ---------------------------------------------------------------------------
# Quantile binning (the columnar-frame equivalent of MLlib's tree binning;
# also backs approxQuantile uses: DummyRegressor.scala:120-124,
# GBMRegressor.scala:306,347-351)
# ---------------------------------------------------------------------------


def quantile_bins(
    x: torch.Tensor, max_bins: int = 256, sample_rows: int = 262_144, seed: int = 17
) -> torch.Tensor:
    """Per-feature quantile cut points.

    x: [N, F] f32 -> edges [F, max_bins-1] f32; bin b covers
    (edges[b-1], edges[b]].  Built from a row sample (cap ``sample_rows``)
    like MLlib's ``findSplits``; duplicated cut points simply leave empty
    bins.
    """
    n, f = x.shape
    if n > sample_rows:
        g, _ = _gen_for(x.device, seed)
        idx = torch.randint(0, n, (sample_rows,), generator=g, device=x.device)
        xs = x.index_select(0, idx)
    else:
        xs = x
    s = xs.shape[0]
    xs_sorted, _ = xs.sort(dim=0)
    # cut at interior quantile positions
    q = torch.arange(1, max_bins, device=x.device, dtype=torch.float32) / max_bins
    pos = (q * (s - 1)).long().clamp_(0, s - 1)
    edges = xs_sorted.index_select(0, pos).T.contiguous()  # [F, B-1]
    return edges


def bin_features(x: torch.Tensor, edges: torch.Tensor) -> torch.Tensor:
    """Map raw features to bin indices: [N, F] f32 -> [N, F] uint8.

    bin = number of edges strictly below-or-equal x (searchsorted right on
    edges gives index of first edge > x ... we use left: x <= edge -> bin of
    that edge).  Convention: row goes LEFT iff bin <= split_bin.
    """
    # searchsorted wants [F, N]
    xt = x.T.contiguous()
    b = torch.searchsorted(edges.contiguous(), xt, right=False)
    return b.T.contiguous().to(torch.uint8)


# ---------------------------------------------------------------------------
# Histogram build (replaces MLlib DecisionTree's treeAggregate histogram
# rounds invoked via fitBaseLearner, reference ensembleParams.scala:64-81)
# ---------------------------------------------------------------------------


def hist_build(
    bins: torch.Tensor,  # [N, F] uint8
    gh: torch.Tensor,  # [N, C] f32  (C = D grad dims + hess + count-weight)
    row_idx: torch.Tensor,  # [M] int32/int64 rows grouped by node
    node_offsets: torch.Tensor,  # [n_nodes+1] int64 segment bounds in row_idx
    num_bins: int,
) -> torch.Tensor:
    """Per-(node, feature, bin) sums of gh channels.

    Returns [n_nodes, F, num_bins, C] f32.  The HIP kernel stages per-
    feature-group histograms in LDS with atomic adds; this reference uses
    index_add_ per node.
    """
    n_nodes = node_offsets.numel() - 1
    N, F = bins.shape
    C = gh.shape[1]
    out = torch.zeros(n_nodes, F, num_bins, C, dtype=torch.float32, device=bins.device)
    offs = node_offsets.tolist()
    fb = torch.arange(F, device=bins.device, dtype=torch.long) * num_bins
    for nd in range(n_nodes):
        s, e = offs[nd], offs[nd + 1]
        if e <= s:
            continue
        rows = row_idx[s:e].long()
        b = bins.index_select(0, rows).long()  # [m, F]
        flat = (b + fb.unsqueeze(0)).reshape(-1)  # [m*F]
        vals = (
            gh.index_select(0, rows)
            .unsqueeze(1)
            .expand(-1, F, -1)
            .reshape(-1, C)
        )
        out[nd].reshape(F * num_bins, C).index_add_(0, flat, vals)
    return out


def hist_build_forest(bins, gh, row_idx, node_offsets, node_col0, num_bins,
                      c_per_node):
    """Forest-build reference: gh [N, T*C] interleaves per-tree channel
    groups; node nd accumulates columns [col0, col0+C) only."""
    n_nodes = node_offsets.numel() - 1
    N, F = bins.shape
    C = int(c_per_node)
    out = torch.zeros(n_nodes, F, num_bins, C, dtype=torch.float32,
                      device=bins.device)
    offs = node_offsets.tolist()
    col0s = node_col0.tolist()
    fb = torch.arange(F, device=bins.device, dtype=torch.long) * num_bins
    for nd in range(n_nodes):
        s, e = offs[nd], offs[nd + 1]
        if e <= s:
            continue
        c0 = int(col0s[nd])
        rows = row_idx[s:e].long()
        b = bins.index_select(0, rows).long()
        flat = (b + fb.unsqueeze(0)).reshape(-1)
        vals = (
            gh[:, c0:c0 + C].index_select(0, rows)
            .unsqueeze(1)
            .expand(-1, F, -1)
            .reshape(-1, C)
        )
        out[nd].reshape(F * num_bins, C).index_add_(0, flat, vals)
    return out


# ---------------------------------------------------------------------------
# Split search (vectorized over nodes x features x bins; works on both
# devices as plain tensor algebra — small relative to hist_build)
# ---------------------------------------------------------------------------


def split_search(
    hist: torch.Tensor,  # [n_nodes, F, B, C]
    lam: float = 1e-6,
    min_child_weight: float = 0.0,
    min_instances: float = 1.0,
    min_info_gain: float = 0.0,
    d_dims: int = -1,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Best variance-reduction / newton-gain split per node.

    Gain for child stats (G vec, H): score = |G|^2 / (H + lam); split gain =
    scoreL + scoreR - scoreParent.  With G = sum of one-hot labels and
    H = count this is exactly weighted gini gain x const; with
    G = residual sums, H = hessian sums it is the XGBoost-style newton gain.

    Returns (gain [n], feature [n], bin [n], left_stats [n, C]).
    gain = -inf where no valid split exists.
    """
    n, F, B, C = hist.shape
    D = d_dims if d_dims > 0 else C - 2
    idx_c = C - 1  # count channel; == hess channel when C == D + 1
    total = hist.sum(dim=2)  # [n, F, B, C] -> [n, F, C]
    parent = total[:, 0, :]  # same for every feature: [n, C]
    cum = hist.cumsum(dim=2)  # left stats if split at bin b (x <= edge_b)
    left = cum[:, :, : B - 1, :]  # last bin can't split
    right = total.unsqueeze(2) - left

    def score(stats):
        g = stats[..., :D]
        h = stats[..., D]
        return (g * g).sum(dim=-1) / (h + lam)

    gain = score(left) + score(right) - score(parent)[:, None, None]
    hl = left[..., D]
    hr = right[..., D]
    cl = left[..., idx_c]
    cr = right[..., idx_c]
    valid = (
        (hl >= min_child_weight)
        & (hr >= min_child_weight)
        & (cl >= min_instances)
        & (cr >= min_instances)
    )
    gain = torch.where(valid, gain, torch.full_like(gain, float("-inf")))
    flat = gain.reshape(n, F * (B - 1))
    best = flat.argmax(dim=1)
    best_gain = flat.gather(1, best.unsqueeze(1)).squeeze(1)
    feat = best // (B - 1)
    b = best % (B - 1)
    left_stats = left[torch.arange(n, device=hist.device), feat, b]  # [n, C]
    # apply min_info_gain
    best_gain = torch.where(
        best_gain >= min_info_gain, best_gain, torch.full_like(best_gain, float("-inf"))
    )
    return best_gain, feat, b, left_stats


# ---------------------------------------------------------------------------
# Row partition (tree growth bookkeeping; MLlib keeps a nodeIdCache — we
# keep explicit per-node row-index segments)
# ---------------------------------------------------------------------------


def partition_rows(
    bins: torch.Tensor,  # [N, F] uint8
    row_idx: torch.Tensor,  # [M]
    node_offsets: torch.Tensor,  # [n+1]
    feat: torch.Tensor,  # [n] best feature per node (-1 = leaf, don't split)
    thr: torch.Tensor,  # [n] split bin per node
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Partition each node's rows into (left, right) by bin <= thr.

    Returns (new_row_idx, new_node_offsets [2n+1], left_counts [n]).
    Nodes with feat < 0 keep all rows on the left side (callers drop them
    from the active set before the next level).
    """
    offs = node_offsets.tolist()
    n = len(offs) - 1
    pieces: List[torch.Tensor] = []
    sizes: List[int] = []
    left_counts = []
    for nd in range(n):
        s, e = offs[nd], offs[nd + 1]
        rows = row_idx[s:e]
        f = int(feat[nd])
        if f < 0:
            pieces.append(rows)
            pieces.append(rows[:0])
            left_counts.append(rows.numel())
            sizes.extend([rows.numel(), 0])
            continue
        go_left = bins[rows.long(), f] <= thr[nd]
        l = rows[go_left]
        r = rows[~go_left]
        pieces.extend([l, r])
        left_counts.append(l.numel())
        sizes.extend([l.numel(), r.numel()])
    new_rows = torch.cat(pieces) if pieces else row_idx[:0]
    new_offs = torch.tensor(
        [0] + list(torch.tensor(sizes).cumsum(0).tolist()), dtype=torch.int64
    )
    return new_rows, new_offs, torch.tensor(left_counts, dtype=torch.int64)


# ---------------------------------------------------------------------------
# Tree inference (reference per-row model.predict loops, e.g.
# BaggingRegressor.scala:221-228, GBMClassifier.scala:567-589)
# ---------------------------------------------------------------------------


def tree_predict(
    x: torch.Tensor,  # [N, F] f32 raw features
    feature: torch.Tensor,  # [n_nodes] int32 (-1 leaf)
    threshold: torch.Tensor,  # [n_nodes] f32 (go left iff x <= thr)
    left_child: torch.Tensor,  # [n_nodes] int32 (right = left+1)
    leaf_value: torch.Tensor,  # [n_nodes, D] f32
    max_depth: int,
) -> torch.Tensor:
    """Vectorized level-synchronous walk; returns [N, D]."""
    N = x.shape[0]
    node = torch.zeros(N, dtype=torch.long, device=x.device)
    featl = feature.long()
    leftl = left_child.long()
    for _ in range(max_depth + 1):
        f = featl[node]
        is_leaf = f < 0
        if bool(is_leaf.all()):
            break
        fx = x.gather(1, f.clamp(min=0).unsqueeze(1)).squeeze(1)
        go_left = fx <= threshold[node]
        nxt = leftl[node] + (~go_left).long()
        node = torch.where(is_leaf, node, nxt)
    return leaf_value[node]


def forest_predict(
    x: torch.Tensor,
    trees: List[dict],
    weights: Optional[torch.Tensor] = None,
    max_depth: int = 64,
) -> torch.Tensor:
    """Sum (optionally weighted) of per-tree predictions: [N, D]."""
    out = None
    for i, t in enumerate(trees):
        p = tree_predict(
            x, t["feature"], t["threshold"], t["left_child"], t["leaf_value"], max_depth
        )
        if weights is not None:
            p = p * weights[i]
        out = p if out is None else out + p
    return out


def logreg_loss_grad(x, y_int, w, wmat, has_bias):
    """Torch reference for the fused logistic loss+grad payload
    (dispatch.logreg_loss_grad contract)."""
    f = x.shape[1]
    k = wmat.shape[1]
    wm = wmat[:f]
    raw = x @ wm
    if has_bias:
        raw = raw + wmat[f]
    logp = torch.log_softmax(raw, dim=1)
    yl = y_int.long()
    nll = -(logp.gather(1, yl.unsqueeze(1)).squeeze(1) * w).sum()
    p = logp.exp()
    onehot = torch.zeros_like(p)
    onehot.scatter_(1, yl.unsqueeze(1), 1.0)
    gmat = (p - onehot) * w.unsqueeze(1)
    gw = x.T @ gmat
    gb = gmat.sum(dim=0) if has_bias else torch.zeros(k, device=x.device)
    return torch.cat([nll.reshape(1), gw.reshape(-1), gb.reshape(-1)])
