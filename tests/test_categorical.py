"""Categorical feature metadata (reference Utils.getFeaturesMetadata,
Utils.scala:42-61: base learners must see categorical feature info, and
the rebuilt AttributeGroup metadata must survive a subspace slice).

MI355X design: categorical features carry integer category ids in the
float features tensor; declared cardinalities switch their binning from
quantile cut points to IDENTITY cut points (bin == category id), so tree
split thresholds are exact category boundaries (ordinal-categorical
handling).  Subspace propagation is free: sliced edge rows travel with
the sliced features (ensemble/binning.py fit_frame)."""

import numpy as np
import pytest
import torch

import spark_ensemble_amd as sea
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.models import DecisionTreeClassifier, DecisionTreeRegressor


def _cat_frame(n=6000, seed=0):
    """Feature 0: categorical with 6 categories (labels depend on it
    non-monotonically), features 1-3: continuous noise."""
    g = torch.Generator().manual_seed(seed)
    cat = torch.randint(0, 6, (n,), generator=g).float()
    noise = torch.randn(n, 3, generator=g)
    x = torch.cat([cat.unsqueeze(1), noise], dim=1)
    y = torch.tensor([1.0, 0.0, 1.0, 0.0, 0.0, 1.0])[cat.long()]
    return TensorFrame(features=x, label=y).set_categorical({0: 6})


def test_categorical_tree_splits_on_exact_category_ids():
    df = _cat_frame()
    m = DecisionTreeClassifier().setMaxDepth(4).setMaxBins(32).fit(df)
    acc = float((m.transform(df)["prediction"] == df["label"]).float().mean())
    assert acc > 0.999, acc
    # every split on feature 0 must sit on an integer category id
    t = m._tree
    on_cat = t["feature"] == 0
    assert bool(on_cat.any())
    thr = t["threshold"][on_cat]
    assert torch.equal(thr, thr.round()), thr
    assert bool((thr >= 0).all() and (thr <= 5).all())


def test_categorical_metadata_survives_frame_transforms():
    df = _cat_frame()
    assert df.categorical == {0: 6}
    assert df.filter(torch.arange(100)).categorical == {0: 6}
    assert df.to("cpu").categorical == {0: 6}
    assert df.withColumn("w", torch.ones(len(df))).categorical == {0: 6}


def test_categorical_through_subspace_slice():
    """A bagged ensemble with subspaceRatio < 1 must keep categorical
    binning for sliced members (the reference rebuilds AttributeGroup
    metadata per subspace — Utils.scala:42-61)."""
    df = _cat_frame(8000)
    bag = (
        sea.BaggingClassifier()
        .setBaseLearner(DecisionTreeClassifier().setMaxDepth(4).setMaxBins(32))
        .setNumBaseLearners(8)
        .setSubspaceRatio(0.5)
        .setVotingStrategy("soft")
        .setSeed(3)
        .fit(df)
    )
    # members that saw feature 0 split it on exact category ids
    saw_cat = 0
    for m, sub in zip(bag._models, bag._subspaces):
        sub_l = sub.tolist()
        if 0 not in sub_l:
            continue
        local_f = sub_l.index(0)
        t = m._tree
        on_cat = t["feature"] == local_f
        if bool(on_cat.any()):
            saw_cat += 1
            thr = t["threshold"][on_cat]
            assert torch.equal(thr, thr.round()), thr
    assert saw_cat > 0
    # members blind to feature 0 (label depends only on it) dilute the
    # vote; soft voting with ~half the members seeing it must still beat
    # the 0.5 majority baseline decisively
    acc = float((bag.transform(df)["prediction"] == df["label"]).float().mean())
    assert acc > 0.8, acc


def test_categorical_cardinality_over_maxbins_raises():
    df = _cat_frame()
    df.set_categorical({0: 300})
    with pytest.raises(ValueError, match="maxBins"):
        DecisionTreeClassifier().setMaxDepth(3).setMaxBins(32).fit(df)


def test_categorical_regression_tree():
    g = torch.Generator().manual_seed(4)
    n = 5000
    cat = torch.randint(0, 5, (n,), generator=g).float()
    x = torch.cat([cat.unsqueeze(1), torch.randn(n, 2, generator=g)], dim=1)
    means = torch.tensor([3.0, -1.0, 7.0, 0.0, -5.0])
    y = means[cat.long()] + 0.01 * torch.randn(n, generator=g)
    df = TensorFrame(features=x, label=y).set_categorical({0: 5})
    m = DecisionTreeRegressor().setMaxDepth(4).fit(df)
    mse = float(((m.predict(x) - y) ** 2).mean())
    assert mse < 0.01, mse


def test_from_pandas_categorical_inference():
    pd = pytest.importorskip("pandas")
    from spark_ensemble_amd.utils.io import from_pandas

    df = pd.DataFrame({
        "color": pd.Categorical(["red", "blue", "green", "red", "blue"] * 20),
        "size": np.arange(100, dtype=np.float64),
        "label": ([0.0, 1.0] * 50),
    })
    fr = from_pandas(df)
    assert fr.categorical == {0: 3}
    assert fr["features"].shape == (100, 2)
    # codes are 0-based category ids
    assert set(fr["features"][:, 0].unique().tolist()) == {0.0, 1.0, 2.0}
