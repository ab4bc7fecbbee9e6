"""Fused forest grower (grow_forest): T trees grown level-synchronously
in shared launches must EXACTLY reproduce T independent grow_tree fits —
same histograms, same splits, same leaves (the MI355X replacement for the
reference's driver-side parallel fit futures,
GBMClassifier.scala:377-411 / BaggingRegressor.scala:145-166)."""

import pytest
import torch

from spark_ensemble_amd.models.tree_grower import (
    GrowParams,
    _grow_tree_seq,
    grow_forest,
    grow_tree,
)
from spark_ensemble_amd.ops import reference


def _data(n=20000, f=12, b=32, seed=0):
    g = torch.Generator().manual_seed(seed)
    x = torch.randn(n, f, generator=g)
    edges = reference.quantile_bins(x, b)
    bins = reference.bin_features(x, edges)
    return x, edges, bins, g


def _assert_tree_equal(a, b, t):
    # structure must match exactly; leaf values only up to the fused
    # column-sum's reduction order (root totals sum [N, T] by column vs a
    # single tree's flat sum -> ~1e-5 relative noise)
    for k in ("feature", "threshold", "left_child"):
        assert torch.equal(a[k].float(), b[k].float()), (t, k, a[k], b[k])
    assert torch.allclose(
        a["leaf_value"], b["leaf_value"], rtol=1e-4, atol=2e-5
    ), (t, a["leaf_value"], b["leaf_value"])


@pytest.mark.parametrize("weighted", [False, True])
def test_forest_matches_independent_trees(weighted):
    n, f, b, T = 20000, 12, 32, 3
    x, edges, bins, g = _data(n, f, b)
    params = GrowParams(max_depth=4, max_bins=b)
    grads = torch.randn(n, T, generator=g)
    hess = (torch.rand(n, generator=g) + 0.5 if weighted
            else torch.ones(n))

    forest_pred = []
    forest = grow_forest(bins, edges, grads, hess, params,
                         train_pred_out=forest_pred)
    assert len(forest) == T

    for t in range(T):
        single_pred = []
        single = _grow_tree_seq(bins, edges, grads[:, t:t + 1].contiguous(),
                           hess, params, train_pred_out=single_pred)
        _assert_tree_equal(forest[t], single, t)
        assert torch.allclose(forest_pred[0][:, t], single_pred[0][:, 0],
                              rtol=1e-5, atol=1e-6)


def test_forest_per_tree_hessians():
    """Newton-style: each tree has its OWN weight column."""
    n, f, b, T = 15000, 8, 32, 4
    x, edges, bins, g = _data(n, f, b, seed=3)
    params = GrowParams(max_depth=3, max_bins=b)
    grads = torch.randn(n, T, generator=g)
    hess = torch.rand(n, T, generator=g) + 0.1

    forest = grow_forest(bins, edges, grads, hess, params)
    for t in range(T):
        single = _grow_tree_seq(bins, edges, grads[:, t:t + 1].contiguous(),
                           hess[:, t].contiguous(), params)
        _assert_tree_equal(forest[t], single, t)


def test_forest_batching_over_max_fused():
    """T above the fused-batch cap splits into batches transparently."""
    from spark_ensemble_amd.models import tree_grower

    n, f, b = 5000, 6, 16
    x, edges, bins, g = _data(n, f, b, seed=5)
    params = GrowParams(max_depth=3, max_bins=b)
    T = 5
    grads = torch.randn(n, T, generator=g)
    hess = torch.ones(n)

    orig = tree_grower.MAX_FUSED_TREES
    tree_grower.MAX_FUSED_TREES = 2
    try:
        pred_b = []
        batched = grow_forest(bins, edges, grads, hess, params,
                              train_pred_out=pred_b)
    finally:
        tree_grower.MAX_FUSED_TREES = orig
    pred_f = []
    full = grow_forest(bins, edges, grads, hess, params,
                       train_pred_out=pred_f)
    for t in range(T):
        _assert_tree_equal(batched[t], full[t], t)
    assert torch.allclose(pred_b[0], pred_f[0])


def test_forest_root_rows_matches_masked_trees():
    """Per-tree root row sets (OOF folds): each fused tree must equal a
    grow_tree fit restricted to the same rows via row_mask."""
    n, f, b, T = 18000, 10, 32, 3
    x, edges, bins, g = _data(n, f, b, seed=7)
    params = GrowParams(max_depth=4, max_bins=b)
    grads = torch.randn(n, T, generator=g)
    hess = torch.ones(n)
    fold = torch.randint(0, T, (n,), generator=g)
    rows = [(fold != t).nonzero(as_tuple=True)[0].to(torch.int32)
            for t in range(T)]

    pred_f: list = []
    forest = grow_forest(bins, edges, grads, hess, params,
                         train_pred_out=pred_f, root_rows=rows)
    for t in range(T):
        single = _grow_tree_seq(bins, edges, grads[:, t:t + 1].contiguous(),
                           hess, params, row_mask=(fold != t))
        _assert_tree_equal(forest[t], single, t)
        # captured training predictions are zero on held-out rows
        held = fold == t
        assert torch.equal(pred_f[0][held, t],
                           torch.zeros(int(held.sum())))


def test_gbm_fit_folds_matches_sequential_oof():
    """GBMRegressor._fit_folds (fused fold-vectorized fit) must
    reproduce num_folds independent weight-masked GBM fits."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.frame import TensorFrame
    from spark_ensemble_amd.utils.io import synthetic_regression

    df = synthetic_regression(9000, 12, seed=23)
    x, y = df["features"], df["label"]
    T = 3
    g = torch.Generator().manual_seed(5)
    fold = torch.randint(0, T, (x.shape[0],), generator=g)

    est = sea.GBMRegressor().setNumBaseLearners(4).setSeed(1)
    assert est._can_fit_folds()
    shared = TensorFrame(features=x, label=y, weight=torch.ones_like(y))
    fused = est._fit_folds(shared, fold, T)

    for t in range(T):
        wmask = (fold != t).float()
        seq = (
            sea.GBMRegressor().setNumBaseLearners(4).setSeed(1)
            .set("weightCol", "weight")
            .fit(TensorFrame(features=x, label=y, weight=wmask))
        )
        # same init/stage-weights up to float (closed-form alpha == the
        # Newton minimizer of the same quadratic)
        assert abs(fused[t]._init._constant - seq._init._constant) < 1e-6
        for wa, wb in zip(fused[t]._weights, seq._weights):
            assert abs(wa - wb) < 1e-5, (wa, wb)
        # early trees identical; late rounds may flip near-tie splits as
        # ~1e-7 stage-weight noise compounds through the margins
        for r in range(2):
            fa, sb = fused[t]._models[r]._tree, seq._models[r]._tree
            assert torch.equal(fa["feature"], sb["feature"]), (t, r)
            assert torch.equal(fa["threshold"], sb["threshold"]), (t, r)
        a = fused[t].predict(x)
        b = seq.predict(x)
        # models must be statistically interchangeable
        held = fold == t
        mse_a = float(((a - y)[held] ** 2).mean())
        mse_b = float(((b - y)[held] ** 2).mean())
        assert abs(mse_a - mse_b) <= 0.02 * max(mse_b, 1e-9), (mse_a, mse_b)
        close = float((a - b).abs().lt(0.05 * y.std()).float().mean())
        assert close > 0.95, close


def test_fused_bagging_matches_sequential():
    """BaggingRegressor's fused path (subspace masks + bag row sets in
    one forest grow) must reproduce the sequential per-member fits."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.models import tree as tree_mod
    from spark_ensemble_amd.utils.io import synthetic_regression

    df = synthetic_regression(12000, 16, seed=31)

    def mk():
        return (
            sea.BaggingRegressor()
            .setNumBaseLearners(6)
            .setSubspaceRatio(0.6)
            .setSubsampleRatio(0.7)
            .setReplacement(True)
            .setSeed(7)
        )

    m_fused = mk().fit(df)

    orig = tree_mod.fit_tree_forest

    def no_fuse(*a, **k):
        raise AssertionError("should not be called")

    # force the sequential branch by making the learner type check fail
    class _DT(tree_mod.DecisionTreeRegressor):
        pass

    m_seq = mk().setBaseLearner(_DT()).fit(df)

    x = df["features"]
    a = m_fused.predict(x)
    b = m_seq.predict(x)
    assert torch.allclose(a, b, rtol=1e-4, atol=1e-5), \
        float((a - b).abs().max())
    # member trees live in the subspace-local feature space (model
    # parity with sliced fits)
    for m, sub in zip(m_fused._models, m_fused._subspaces):
        f = m._tree["feature"]
        assert int(f.max()) < int(sub.numel())
        assert m._num_features == int(sub.numel())


def test_fused_classification_bagging_matches_sequential():
    """BaggingClassifier's fused gini-forest path (multi-output trees,
    D = K channels) must reproduce sequential member fits."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.models import tree as tree_mod
    from spark_ensemble_amd.utils.io import synthetic_classification

    df = synthetic_classification(12000, 14, k=3, seed=19)

    def mk():
        return (
            sea.BaggingClassifier()
            .setNumBaseLearners(5)
            .setSubspaceRatio(0.6)
            .setSubsampleRatio(0.7)
            .setReplacement(True)
            .setVotingStrategy("soft")
            .setSeed(3)
        )

    m_fused = mk().fit(df)

    class _DT(tree_mod.DecisionTreeClassifier):
        pass  # type check fails -> sequential path

    m_seq = mk().setBaseLearner(_DT()).fit(df)
    a = m_fused.transform(df)["probability"]
    b = m_seq.transform(df)["probability"]
    assert torch.allclose(a, b, rtol=1e-4, atol=1e-5), \
        float((a - b).abs().max())
    for m, sub in zip(m_fused._models, m_fused._subspaces):
        assert int(m._tree["feature"].max()) < int(sub.numel())
        assert m._num_classes == 3


def test_gbm_classifier_fit_folds_matches_sequential():
    """GBMClassifier._fit_folds: fused (fold x class) forests + per-fold
    stage searches must reproduce sequential weight-masked fits."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.frame import TensorFrame
    from spark_ensemble_amd.utils.io import synthetic_classification

    df = synthetic_classification(9000, 12, k=3, seed=29)
    x, y = df["features"], df["label"]
    T = 3
    g = torch.Generator().manual_seed(8)
    fold = torch.randint(0, T, (x.shape[0],), generator=g)

    est = sea.GBMClassifier().setNumBaseLearners(3).setSeed(4)
    assert est._can_fit_folds()
    shared = TensorFrame(features=x, label=y, weight=torch.ones_like(y))
    fused, final = est._fit_folds(shared, fold, T, include_full=True)
    assert len(fused) == T

    for t in range(T):
        wmask = (fold != t).float()
        seq = (
            sea.GBMClassifier().setNumBaseLearners(3).setSeed(4)
            .set("weightCol", "weight")
            .fit(TensorFrame(features=x, label=y, weight=wmask))
        )
        for wa_row, wb_row in zip(fused[t]._weights, seq._weights):
            for wa, wb in zip(wa_row, wb_row):
                assert abs(wa - wb) < 1e-4, (wa, wb)
        a = fused[t].transform(df)["probability"]
        b = seq.transform(df)["probability"]
        close = float((a - b).abs().lt(0.02).float().mean())
        assert close > 0.97, close
        held = fold == t
        acc_a = float((fused[t].transform(df)["prediction"][held]
                       == y[held]).float().mean())
        acc_b = float((seq.transform(df)["prediction"][held]
                       == y[held]).float().mean())
        assert abs(acc_a - acc_b) < 0.02, (acc_a, acc_b)

    # the rode-along full refit equals a plain fit
    seq_full = (
        sea.GBMClassifier().setNumBaseLearners(3).setSeed(4)
        .set("weightCol", "weight")
        .fit(TensorFrame(features=x, label=y, weight=torch.ones_like(y)))
    )
    a = final.transform(df)["probability"]
    b = seq_full.transform(df)["probability"]
    assert torch.allclose(a, b, rtol=1e-3, atol=1e-3), \
        float((a - b).abs().max())


def test_bagging_fit_folds_matches_sequential():
    """BaggingRegressor._fit_folds (all fold x member trees in one
    forest) must reproduce per-fold weight-masked fits."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.frame import TensorFrame
    from spark_ensemble_amd.utils.io import synthetic_regression

    df = synthetic_regression(10000, 14, seed=37)
    x, y = df["features"], df["label"]
    T = 3
    g = torch.Generator().manual_seed(2)
    fold = torch.randint(0, T, (x.shape[0],), generator=g)

    def mk():
        return (
            sea.BaggingRegressor()
            .setNumBaseLearners(4)
            .setSubspaceRatio(0.6)
            .setSubsampleRatio(0.8)
            .setReplacement(True)
            .setSeed(9)
        )

    est = mk()
    assert est._can_fit_folds()
    shared = TensorFrame(features=x, label=y, weight=torch.ones_like(y))
    fused = est._fit_folds(shared, fold, T)

    for f in range(T):
        wmask = (fold != f).float()
        seq = mk().set("weightCol", "weight").fit(
            TensorFrame(features=x, label=y, weight=wmask)
        )
        a = fused[f].predict(x)
        b = seq.predict(x)
        assert torch.allclose(a, b, rtol=1e-4, atol=1e-5), \
            (f, float((a - b).abs().max()))


def test_crossvalidator_fused_folds_matches_sequential():
    """CV routes fold-vectorizable estimators through _fit_folds; scores
    may differ slightly from subset-refit folds (fused folds share
    global cut points) but ranking must agree on a separated grid."""
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.tuning import (
        CrossValidator,
        RegressionEvaluator,
    )
    from spark_ensemble_amd.utils.io import synthetic_regression

    df = synthetic_regression(8000, 10, seed=3)
    grid = [{"learningRate": 1.0, "numBaseLearners": 5},
            {"learningRate": 0.05, "numBaseLearners": 2}]

    def run(fused):
        est = sea.GBMRegressor().setSeed(2)
        if not fused:
            est._can_fit_folds = lambda: False
        cv = CrossValidator(estimator=est, estimatorParamMaps=grid,
                            evaluator=RegressionEvaluator("rmse"),
                            numFolds=3, seed=5)
        return cv.fit(df)

    a = run(True)
    b = run(False)
    assert a.bestIndex == b.bestIndex == 0
    for ma, mb in zip(a.avgMetrics, b.avgMetrics):
        assert abs(ma - mb) / max(abs(mb), 1e-9) < 0.1, (ma, mb)


def test_hist_build_forest_reference():
    """The per-node column-offset histogram itself."""
    g = torch.Generator().manual_seed(9)
    n, f, b, T, C = 8000, 5, 16, 3, 2
    bins = torch.randint(0, b, (n, f), generator=g, dtype=torch.uint8)
    gh = torch.randn(n, T * C, generator=g)
    gh[:, 1::C] = gh[:, 1::C].abs()  # nonneg weight slots
    row_idx = torch.arange(n, dtype=torch.int32).repeat(2)
    offsets = torch.tensor([0, n, n + 4000, 2 * n])
    node_col0 = torch.tensor([0, 2, 4], dtype=torch.int32)
    got = reference.hist_build_forest(bins, gh, row_idx, offsets,
                                      node_col0, b, C)
    for nd, c0 in enumerate(node_col0.tolist()):
        want = reference.hist_build(
            bins, gh[:, c0:c0 + C].contiguous(),
            row_idx[offsets[nd]:offsets[nd + 1]],
            torch.tensor([0, int(offsets[nd + 1] - offsets[nd])]), b,
        )
        assert torch.allclose(got[nd], want[0], atol=1e-4)


@pytest.mark.parametrize("masked", [False, True])
def test_grow_tree_delegation_matches_loop_grower(masked):
    """grow_tree (T=1 grow_forest delegation) == the loop implementation,
    with and without a zero-weight row mask."""
    n, f, b = 15000, 10, 32
    x, edges, bins, g = _data(n, f, b, seed=7)
    params = GrowParams(max_depth=5, max_bins=b)
    grad = torch.randn(n, 1, generator=g)
    hess = torch.ones(n)
    mask = (torch.rand(n, generator=g) > 0.3) if masked else None

    tp_new, tp_old = [], []
    new = grow_tree(bins, edges, grad, hess, params, row_mask=mask,
                    train_pred_out=tp_new)
    old = _grow_tree_seq(bins, edges, grad, hess, params, row_mask=mask,
                         train_pred_out=tp_old)
    _assert_tree_equal(new, old, "delegated")
    if not masked:
        # same tolerance as the leaf values (fused column-sum order)
        assert torch.allclose(tp_new[0], tp_old[0], rtol=1e-4, atol=2e-5)


def test_grow_tree_delegation_multioutput():
    """D>1 (gini) delegation parity, plus the wide-D fallback path."""
    n, f, b, D = 12000, 8, 32, 3
    x, edges, bins, g = _data(n, f, b, seed=11)
    params = GrowParams(max_depth=4, max_bins=b)
    onehot = torch.zeros(n, D)
    onehot[torch.arange(n), torch.randint(0, D, (n,), generator=g)] = 1.0
    hess = torch.ones(n)
    new = grow_tree(bins, edges, onehot, hess, params)
    old = _grow_tree_seq(bins, edges, onehot, hess, params)
    _assert_tree_equal(new, old, "gini")
