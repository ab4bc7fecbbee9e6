"""Property-based tests (hypothesis) — the MI355X analog of the
reference's ScalaCheck suites (HasSubBagSuite.scala:60-105 forAll
properties, UtilsSuite.scala:29-67, GBMLossSuite gradient checks)."""

import os

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from spark_ensemble_amd.ensemble.utils import subspace, weighted_median
from spark_ensemble_amd.boosting.losses import (
    BernoulliLoss, ExponentialLoss, HuberLoss, LogCoshLoss, LogLoss,
    QuantileLoss, ScaledLogCoshLoss, SquaredLoss,
)


@settings(max_examples=40, deadline=None)
@given(
    ratio=st.floats(0.05, 1.0),
    nf=st.integers(1, 300),
    seed=st.integers(0, 2**31 - 1),
)
def test_subspace_forall(ratio, nf, seed):
    idx = subspace(ratio, nf, seed)
    # sorted unique indices in range (reference: sorted output of a
    # Bernoulli(ratio) filter, HasSubBag.scala:73-79)
    assert (idx[1:] > idx[:-1]).all()
    assert idx.numel() >= 1
    assert 0 <= int(idx.min()) and int(idx.max()) < nf
    # deterministic in seed
    assert torch.equal(idx, subspace(ratio, nf, seed))
    if ratio == 1.0:
        assert idx.numel() == nf


@settings(max_examples=30, deadline=None)
@given(
    n=st.integers(1, 200),
    seed=st.integers(0, 10_000),
    scale=st.floats(0.1, 100.0),
)
def test_weighted_median_forall(n, seed, scale):
    g = torch.Generator().manual_seed(seed)
    v = torch.randn(1, n, generator=g)
    w = torch.rand(1, n, generator=g) + 0.01
    m = float(weighted_median(v, w))
    # scaling weights leaves the weighted median unchanged
    m2 = float(weighted_median(v, w * scale))
    assert abs(m - m2) < 1e-5
    # the weighted median is one of the values and satisfies the
    # cumulative-weight >= 50% rule
    order = torch.argsort(v[0])
    vs, ws = v[0][order], w[0][order]
    cum = torch.cumsum(ws, 0) / ws.sum()
    k = int((cum >= 0.5).nonzero()[0])
    assert abs(m - float(vs[k])) < 1e-6


LOSSES = [
    SquaredLoss(), LogCoshLoss(), ScaledLogCoshLoss(0.3), HuberLoss(1.1),
    QuantileLoss(0.7), BernoulliLoss(), ExponentialLoss(),
]


@settings(max_examples=25, deadline=None)
@given(
    li=st.integers(0, len(LOSSES) - 1),
    seed=st.integers(0, 10_000),
)
def test_loss_gradient_finite_difference_forall(li, seed):
    """Gradient check vs central finite differences (the reference runs
    Breeze GradientTester over every loss — GBMLossSuite.scala:84-125)."""
    loss = LOSSES[li]
    g = torch.Generator().manual_seed(seed)
    n = 64
    if loss.name in ("bernoulli", "exponential"):
        y = loss.encode_label(torch.randint(0, 2, (n,), generator=g).float())
    else:
        y = torch.randn(n, generator=g)
    y = y.reshape(n, 1).double()
    p = (torch.randn(n, 1, generator=g) * 0.7).double()
    eps = 1e-6
    num = ((loss.loss(y, p + eps) - loss.loss(y, p - eps)) / (2 * eps)).reshape(-1)
    ana = loss.gradient(y, p).reshape(-1)
    resid = (y - p).reshape(-1)
    # skip rows at a kink (huber delta boundary, quantile zero residual)
    ok = torch.isfinite(num)
    if loss.name == "huber":
        ok &= ((resid.abs() - loss.param).abs() > 1e-4)
    if loss.name in ("quantile", "absolute"):
        ok &= (resid.abs() > 1e-4)
    assert torch.allclose(num[ok], ana[ok], rtol=1e-4, atol=1e-5), loss.name


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(0, 10_000), k=st.integers(2, 6))
def test_logloss_gradient_forall(seed, k):
    loss = LogLoss(k)
    g = torch.Generator().manual_seed(seed)
    n = 32
    y = loss.encode_label(torch.randint(0, k, (n,), generator=g).float()).double()
    p = (torch.randn(n, k, generator=g) * 0.5).double()
    eps = 1e-6
    ana = loss.gradient(y, p)
    for d in range(k):
        dp = torch.zeros_like(p)
        dp[:, d] = eps
        num = ((loss.loss(y, p + dp) - loss.loss(y, p - dp)) / (2 * eps)).reshape(-1)
        assert torch.allclose(num, ana[:, d].reshape(-1), rtol=1e-4, atol=1e-5)


@settings(max_examples=15, deadline=None)
@given(
    n=st.integers(200, 4000),
    f=st.integers(2, 12),
    b=st.sampled_from([8, 16, 32]),
    depth=st.integers(1, 6),
    weighted=st.booleans(),
    masked=st.booleans(),
    seed=st.integers(0, 2**31 - 1),
)
def test_grow_tree_delegation_forall(n, f, b, depth, weighted, masked, seed):
    """The arena grower (grow_forest T=1) and the loop grower agree on
    ARBITRARY shapes, depths, weightings and row masks — the fuzzed
    version of the fixed-shape delegation parity tests."""
    from spark_ensemble_amd.models.tree_grower import (
        GrowParams, _grow_tree_seq, grow_tree,
    )
    from spark_ensemble_amd.ops import reference

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, f, generator=g)
    edges = reference.quantile_bins(x, b)
    bins = reference.bin_features(x, edges)
    grad = torch.randn(n, 1, generator=g)
    hess = (torch.rand(n, generator=g) + 0.5 if weighted
            else torch.ones(n))
    mask = (torch.rand(n, generator=g) > 0.25) if masked else None
    if mask is not None and int(mask.sum()) == 0:
        return
    params = GrowParams(max_depth=depth, max_bins=b)
    a = grow_tree(bins, edges, grad, hess, params, row_mask=mask)
    c = _grow_tree_seq(bins, edges, grad, hess, params, row_mask=mask)
    for k in ("feature", "threshold", "left_child"):
        assert torch.equal(a[k].float(), c[k].float()), k
    assert torch.allclose(a["leaf_value"], c["leaf_value"],
                          rtol=1e-4, atol=2e-5)


@settings(max_examples=8, deadline=None)
@given(
    algo=st.sampled_from(["gbm_reg", "gbm_clf", "bag_clf", "boost_reg"]),
    k=st.integers(2, 3),
    depth=st.integers(2, 4),
    lr=st.floats(0.1, 1.0),
    seed=st.integers(0, 10_000),
)
def test_save_load_roundtrip_forall(tmp_path_factory, algo, k, depth, lr,
                                    seed):
    """Arbitrary estimator configs survive save -> load with identical
    predictions AND identical param values (reference
    DefaultParamsReader/Writer contract across every suite)."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd import persistence
    from spark_ensemble_amd.frame import TensorFrame
    from spark_ensemble_amd.models import DecisionTreeRegressor

    g = torch.Generator().manual_seed(seed)
    x = torch.randn(800, 6, generator=g)
    yr = x[:, 0] - 0.4 * x[:, 1] + 0.1 * torch.randn(800, generator=g)
    yc = (torch.rand(800, generator=g) * k).floor().clamp(0, k - 1)

    base = DecisionTreeRegressor().setMaxDepth(depth)
    if algo == "gbm_reg":
        est = (sea.GBMRegressor().setNumBaseLearners(2).setBaseLearner(base)
               .setLearningRate(lr).setSeed(seed))
        df = TensorFrame(features=x, label=yr)
    elif algo == "gbm_clf":
        est = (sea.GBMClassifier().setNumBaseLearners(2).setBaseLearner(base)
               .setLearningRate(lr).setSeed(seed))
        df = TensorFrame(features=x, label=yc)
    elif algo == "bag_clf":
        est = (sea.BaggingClassifier().setNumBaseLearners(3)
               .setSubspaceRatio(0.8).setSeed(seed))
        df = TensorFrame(features=x, label=yc)
    else:
        est = sea.BoostingRegressor().setNumBaseLearners(2).setSeed(seed)
        df = TensorFrame(features=x, label=yr)

    model = est.fit(df)
    p = os.path.join(str(tmp_path_factory.mktemp("rt")), "m")
    model.save(p)
    loaded = persistence.load_instance(p)
    a = model.transform(df)["prediction"]
    b = loaded.transform(df)["prediction"]
    assert torch.allclose(a, b, rtol=1e-6, atol=1e-7)
