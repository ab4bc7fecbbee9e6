"""Degenerate-input robustness (the reference's genre: zero-error stop
BoostingRegressorSuite.scala:154-167, wrong-column throws :169-182 —
extended here to tiny/constant/NaN/single-class inputs)."""

import torch

import spark_ensemble_amd as sea
from spark_ensemble_amd.frame import TensorFrame


def test_constant_feature_is_never_split():
    x = torch.ones(100, 3)
    x[:, 1] = torch.randn(100, generator=torch.Generator().manual_seed(1))
    y = (x[:, 1] > 0).float()
    m = sea.DecisionTreeClassifier().setMaxDepth(3).fit(
        TensorFrame(features=x, label=y)
    )
    acc = float((m.transform(TensorFrame(features=x, label=y))["prediction"] == y)
                .float().mean())
    assert acc > 0.95
    feats = m._tree["feature"]
    assert not bool((feats == 0).any()) and not bool((feats == 2).any())


def test_single_row_fit():
    df = TensorFrame(features=torch.randn(1, 4), label=torch.tensor([2.5]))
    m = sea.DecisionTreeRegressor().fit(df)
    assert abs(float(m.predict(df["features"])[0]) - 2.5) < 1e-5


def test_single_class_gbm_classifier():
    df = TensorFrame(features=torch.randn(50, 4), label=torch.zeros(50))
    m = sea.GBMClassifier().setNumBaseLearners(2).fit(df)
    out = m.transform(df)
    assert bool((out["prediction"] == 0).all())


def test_nan_feature_values_bin_consistently():
    g = torch.Generator().manual_seed(3)
    x = torch.randn(200, 4, generator=g)
    x[5, 2] = float("nan")
    y = (x[:, 0] > 0).float()
    df = TensorFrame(features=x, label=y)
    m = sea.DecisionTreeClassifier().fit(df)
    p = m.transform(df)["prediction"]
    assert bool(torch.isfinite(p).all())


def test_separable_data_stops_boosting_early():
    x = torch.cat([torch.zeros(20, 2), torch.ones(20, 2)])
    y = torch.cat([torch.zeros(20), torch.ones(20)])
    m = sea.BoostingClassifier().setNumBaseLearners(10).fit(
        TensorFrame(features=x, label=y)
    )
    # perfect learner => est_err <= 0 => done after one round
    assert m.numModels == 1
    out = m.transform(TensorFrame(features=x, label=y))
    assert bool((out["prediction"] == y).all())


def test_more_bins_than_distinct_values():
    x = torch.tensor([[0.0], [1.0], [2.0]] * 30)
    y = x[:, 0].clone()
    m = sea.DecisionTreeRegressor().setMaxBins(256).fit(
        TensorFrame(features=x, label=y)
    )
    p = m.predict(x)
    assert torch.allclose(p, y, atol=1e-5)


def test_feature_width_mismatch_raises():
    import pytest

    df = TensorFrame(features=torch.randn(100, 5),
                     label=torch.randint(0, 2, (100,)).float())
    m = sea.GBMClassifier().setNumBaseLearners(2).fit(df)
    bad = TensorFrame(features=torch.randn(10, 7))
    with pytest.raises(ValueError, match="trained on 5 features"):
        m.transform(bad)
