"""Unit tests of the compute ops (torch reference path) — these same
semantics are re-checked against the HIP kernels in test_gpu.py."""

import torch

from spark_ensemble_amd.ops import reference as ops


def test_sample_weights_poisson_mean():
    w = ops.sample_weights(True, 0.7, 200_000, seed=3)
    assert abs(float(w.mean()) - 0.7) < 0.02
    assert (w == w.round()).all()  # integer multiplicities


def test_sample_weights_bernoulli():
    w = ops.sample_weights(False, 0.3, 200_000, seed=4)
    assert set(w.unique().tolist()) <= {0.0, 1.0}
    assert abs(float(w.mean()) - 0.3) < 0.02


def test_sample_weights_deterministic_in_seed():
    a = ops.sample_weights(True, 0.5, 1000, seed=9)
    b = ops.sample_weights(True, 0.5, 1000, seed=9)
    c = ops.sample_weights(True, 0.5, 1000, seed=10)
    assert torch.equal(a, b)
    assert not torch.equal(a, c)


def test_binning_roundtrip_rule():
    g = torch.Generator().manual_seed(1)
    x = torch.randn(5000, 3, generator=g)
    edges = ops.quantile_bins(x, 16)
    bins = ops.bin_features(x, edges)
    assert bins.dtype == torch.uint8
    assert int(bins.max()) <= 15
    # invariant: bin b  <=>  (edges[b-1] < x <= edges[b]) with edges[-1]=-inf
    f = 1
    e = edges[f]
    xb = x[:, f]
    b = bins[:, f].long()
    inner = (b > 0) & (b < 15)
    assert bool((xb[inner] <= e[b[inner]]).all())
    assert bool((xb[inner] > e[b[inner] - 1]).all())
    # approximately equal occupancy for continuous data
    counts = torch.bincount(bins[:, 0].long(), minlength=16).float()
    assert float(counts.std() / counts.mean()) < 0.2


def test_hist_build_matches_naive():
    g = torch.Generator().manual_seed(2)
    n, f, b, d = 500, 6, 8, 2
    bins = torch.randint(0, b, (n, f), generator=g, dtype=torch.uint8)
    gh = torch.randn(n, d + 2, generator=g)
    rows = torch.randperm(n, generator=g)[:400].to(torch.int32)
    offs = torch.tensor([0, 150, 150, 400])  # includes an empty node
    out = ops.hist_build(bins, gh, rows, offs, b)
    assert out.shape == (3, f, b, d + 2)
    # naive recompute for node 2
    naive = torch.zeros(f, b, d + 2)
    for r in rows[150:400].tolist():
        for fi in range(f):
            naive[fi, int(bins[r, fi])] += gh[r]
    assert torch.allclose(out[2], naive, atol=1e-4)
    assert out[1].abs().sum() == 0


def test_split_search_finds_planted_split():
    # single feature, clean separation at bin <= 3 -> gain must pick it
    n, f, b = 400, 3, 8
    g = torch.Generator().manual_seed(3)
    bins = torch.randint(0, b, (n, f), generator=g, dtype=torch.uint8)
    y = (bins[:, 1] <= 3).float() * 2.0 - 1.0
    gh = torch.stack([y, torch.ones(n), torch.ones(n)], dim=1)
    hist = ops.hist_build(
        bins, gh, torch.arange(n, dtype=torch.int32), torch.tensor([0, n]), b
    )
    gain, feat, bsplit, left = ops.split_search(hist)
    assert int(feat[0]) == 1
    assert int(bsplit[0]) == 3
    assert float(left[0, 0]) == float(y[bins[:, 1] <= 3].sum())


def test_split_search_respects_min_instances():
    n, b = 100, 4
    bins = torch.zeros(n, 1, dtype=torch.uint8)
    bins[:2, 0] = 0
    bins[2:, 0] = 3
    y = torch.ones(n)
    y[:2] = -1
    gh = torch.stack([y, torch.ones(n), torch.ones(n)], dim=1)
    hist = ops.hist_build(
        bins, gh, torch.arange(n, dtype=torch.int32), torch.tensor([0, n]), b
    )
    gain, _, _, _ = ops.split_search(hist, min_instances=5)
    assert not torch.isfinite(gain[0])


def test_partition_rows():
    g = torch.Generator().manual_seed(4)
    n, f, b = 300, 4, 8
    bins = torch.randint(0, b, (n, f), generator=g, dtype=torch.uint8)
    rows = torch.arange(n, dtype=torch.int32)
    offs = torch.tensor([0, n])
    feat = torch.tensor([2], dtype=torch.int32)
    thr = torch.tensor([4], dtype=torch.int32)
    new_rows, new_offs, lc = ops.partition_rows(bins, rows, offs, feat, thr)
    l = new_rows[: new_offs[1]]
    r = new_rows[new_offs[1] : new_offs[2]]
    assert (bins[l.long(), 2] <= 4).all()
    assert (bins[r.long(), 2] > 4).all()
    assert l.numel() + r.numel() == n
    assert int(lc[0]) == l.numel()


def test_tree_predict_walks_correctly():
    # hand-built depth-2 tree: root on f0<=0.5, left leaf 1.0,
    # right splits f1<=0.0 -> 2.0 / 3.0
    feature = torch.tensor([0, -1, 1, -1, -1], dtype=torch.int32)
    threshold = torch.tensor([0.5, 0.0, 0.0, 0.0, 0.0])
    left = torch.tensor([1, -1, 3, -1, -1], dtype=torch.int32)
    leaf = torch.tensor([[0.0], [1.0], [0.0], [2.0], [3.0]])
    x = torch.tensor([[0.2, 9.0], [0.9, -1.0], [0.9, 1.0], [0.5, 5.0]])
    out = ops.tree_predict(x, feature, threshold, left, leaf, 4)
    assert out.squeeze(1).tolist() == [1.0, 2.0, 3.0, 1.0]


def test_forest_predict_weighted_sum():
    feature = torch.tensor([-1], dtype=torch.int32)
    threshold = torch.tensor([0.0])
    left = torch.tensor([-1], dtype=torch.int32)
    t1 = {
        "feature": feature, "threshold": threshold, "left_child": left,
        "leaf_value": torch.tensor([[2.0]]),
    }
    t2 = {
        "feature": feature, "threshold": threshold, "left_child": left,
        "leaf_value": torch.tensor([[5.0]]),
    }
    x = torch.zeros(3, 1)
    out = ops.forest_predict(x, [t1, t2], torch.tensor([1.0, 0.5]))
    assert out.squeeze(1).tolist() == [4.5, 4.5, 4.5]


def test_gather_ranges_cpu_fallback():
    import torch as t
    from spark_ensemble_amd.ops import dispatch

    src = t.arange(100, dtype=t.int32)
    starts = t.tensor([5, 40, 90], dtype=t.int64)
    lens = t.tensor([3, 0, 10], dtype=t.int64)
    out = dispatch.gather_ranges(src, starts, lens)
    assert out.tolist() == list(range(5, 8)) + list(range(90, 100))


def test_leaf_scatter_cpu_fallback():
    import torch as t
    from spark_ensemble_amd.ops import dispatch

    tp = t.zeros(10, 2)
    row_idx = t.tensor([3, 4, 5, 9, 0], dtype=t.int32)
    dispatch.leaf_scatter(
        tp, row_idx,
        t.tensor([0, 3], dtype=t.int64), t.tensor([3, 2], dtype=t.int64),
        t.tensor([0, 1], dtype=t.int64), t.tensor([2.5, -1.0]),
    )
    assert tp[3, 0] == 2.5 and tp[4, 0] == 2.5 and tp[5, 0] == 2.5
    assert tp[9, 1] == -1.0 and tp[0, 1] == -1.0
    assert tp.abs().sum() == 2.5 * 3 + 2.0
