"""GBM statistical suites (reference GBMClassifierSuite / GBMRegressorSuite:
beats base learner and plain boosting, monotone improvement with more
learners, EXACT early-stop index vs an offline patience-rule replay,
round-trips incl. the exponential loss)."""

import pytest
import torch

from spark_ensemble_amd import (
    GBMClassificationModel,
    GBMClassifier,
    GBMRegressionModel,
    GBMRegressor,
)
from spark_ensemble_amd.boosting.losses import get_classification_loss, get_regression_loss
from spark_ensemble_amd.ensemble.utils import slice_features
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.models import DecisionTreeClassifier, DecisionTreeRegressor


def _acc(model, frame):
    out = model.transform(frame)
    return float((out["prediction"] == frame["label"]).float().mean())


def _rmse(model, frame):
    return float(((model.predict(frame["features"]) - frame["label"]) ** 2).mean() ** 0.5)


def test_gbm_classifier_beats_tree(clf_frame, clf_frame_test):
    single = DecisionTreeClassifier().setMaxDepth(5).fit(clf_frame)
    gbm = GBMClassifier().setNumBaseLearners(10).fit(clf_frame)
    assert _acc(gbm, clf_frame_test) > _acc(single, clf_frame_test)


def test_gbm_regressor_beats_tree(reg_frame, reg_frame_test):
    single = DecisionTreeRegressor().setMaxDepth(5).fit(reg_frame)
    gbm = GBMRegressor().setNumBaseLearners(15).fit(reg_frame)
    assert _rmse(gbm, reg_frame_test) < _rmse(single, reg_frame_test)


def test_gbm_monotone_improvement(reg_frame, reg_frame_test):
    gbm = GBMRegressor().setNumBaseLearners(15).setLearningRate(0.5).fit(reg_frame)

    def prefix_rmse(k):
        m = GBMRegressionModel()
        m._init = gbm._init
        m._models = gbm._models[:k]
        m._weights = gbm._weights[:k]
        m._subspaces = gbm._subspaces[:k]
        for p in ("featuresCol", "predictionCol", "labelCol"):
            m.set(p, gbm.getOrDefault(p))
        return _rmse(m, reg_frame_test)

    # reference GBMRegressorSuite:126-164 (prefix-model evaluation)
    assert prefix_rmse(15) < prefix_rmse(3) < prefix_rmse(1)


@pytest.mark.parametrize("loss", ["squared", "absolute", "huber", "quantile", "logcosh"])
def test_gbm_regressor_losses_run(reg_frame, loss):
    gbm = (
        GBMRegressor()
        .setLoss(loss)
        .setNumBaseLearners(3)
        .setAlpha(0.5)
        .fit(reg_frame)
    )
    assert len(gbm._models) == 3
    assert torch.isfinite(gbm.predict(reg_frame["features"])).all()


@pytest.mark.parametrize("loss", ["logloss", "bernoulli", "exponential"])
def test_gbm_classifier_losses_run(bin_frame, bin_frame_test, loss):
    gbm = GBMClassifier().setLoss(loss).setNumBaseLearners(5).fit(bin_frame)
    acc = _acc(gbm, bin_frame_test)
    assert acc > 0.6, (loss, acc)
    prob = gbm.transform(bin_frame_test)["probability"]
    assert torch.allclose(prob.sum(dim=1), torch.ones_like(prob[:, 0]), atol=1e-5)


def test_gbm_newton_updates(bin_frame, bin_frame_test):
    gbm = (
        GBMClassifier()
        .setLoss("bernoulli")
        .setUpdates("newton")
        .setNumBaseLearners(5)
        .fit(bin_frame)
    )
    assert _acc(gbm, bin_frame_test) > 0.6


def test_gbm_multiclass_fits_k_models_per_round(clf_frame):
    gbm = GBMClassifier().setLoss("logloss").setNumBaseLearners(3).fit(clf_frame)
    assert len(gbm._models) == 3
    assert len(gbm._models[0]) == 3  # K = 3 classes -> dim = 3


def test_gbm_subbagging(reg_frame, reg_frame_test):
    gbm = (
        GBMRegressor()
        .setNumBaseLearners(10)
        .setSubsampleRatio(0.6)
        .setSubspaceRatio(0.7)
        .fit(reg_frame)
    )
    single = DecisionTreeRegressor().setMaxDepth(5).fit(reg_frame)
    assert _rmse(gbm, reg_frame_test) < _rmse(single, reg_frame_test) * 1.1


def _offline_early_stop_replay(gbm_full, xv, yv, loss, validation_tol, num_rounds):
    """Replay the reference patience rule (GBMRegressor.scala:444-465) over
    the full stage list; returns the kept stage count i - v."""
    pred = gbm_full._init.predict(xv)
    yv2 = yv.unsqueeze(1)
    best = float(loss.loss(yv2, pred.unsqueeze(1)).mean())
    v = 0
    i = 0
    for w, sub, m in zip(gbm_full._weights, gbm_full._subspaces, gbm_full._models):
        pred = pred + w * m.predict(slice_features(xv, sub))
        err = float(loss.loss(yv2, pred.unsqueeze(1)).mean())
        i += 1
        if best - err < validation_tol * max(err, 0.01):
            v += 1
            if v >= num_rounds:
                break
        elif err < best:
            best = err
            v = 0
    return i - v


def test_gbm_early_stop_index_exact(reg_frame):
    # build a validation split column
    n = reg_frame.count()
    g = torch.Generator().manual_seed(77)
    is_val = (torch.rand(n, generator=g) < 0.3).float()
    df = reg_frame.withColumn("isVal", is_val)

    full = GBMRegressor().setNumBaseLearners(12).setSeed(5).fit(
        df.filter(~is_val.bool())
    )
    stopped = (
        GBMRegressor()
        .setNumBaseLearners(12)
        .setSeed(5)
        .setValidationIndicatorCol("isVal")
        .setValidationTol(0.01)
        .setNumRounds(1)
        .fit(df)
    )
    xv = df.filter(is_val.bool())["features"]
    yv = df.filter(is_val.bool())["label"]
    loss = get_regression_loss("squared")
    expected = _offline_early_stop_replay(full, xv, yv, loss, 0.01, 1)
    assert stopped.numModels == expected


def test_gbm_classifier_roundtrip(tmp_path, clf_frame):
    gbm = GBMClassifier().setNumBaseLearners(3).fit(clf_frame)
    p = str(tmp_path / "g")
    gbm.save(p)
    loaded = GBMClassificationModel.load(p)
    o1 = gbm.transform(clf_frame)
    o2 = loaded.transform(clf_frame)
    assert torch.equal(o1["prediction"], o2["prediction"])
    assert torch.allclose(o1["rawPrediction"], o2["rawPrediction"], atol=1e-5)


def test_gbm_classifier_exponential_roundtrip(tmp_path, bin_frame):
    gbm = (
        GBMClassifier().setLoss("exponential").setNumBaseLearners(3).fit(bin_frame)
    )
    p = str(tmp_path / "ge")
    gbm.save(p)
    loaded = GBMClassificationModel.load(p)
    assert torch.equal(
        gbm.transform(bin_frame)["prediction"],
        loaded.transform(bin_frame)["prediction"],
    )


def test_gbm_regressor_roundtrip(tmp_path, reg_frame):
    gbm = GBMRegressor().setNumBaseLearners(3).setSubspaceRatio(0.8).fit(reg_frame)
    p = str(tmp_path / "gr")
    gbm.save(p)
    loaded = GBMRegressionModel.load(p)
    assert torch.allclose(
        gbm.predict(reg_frame["features"]),
        loaded.predict(reg_frame["features"]),
        atol=1e-6,
    )


def test_gbm_init_strategies(reg_frame):
    for strat in ("constant", "zero", "base"):
        gbm = GBMRegressor().setInitStrategy(strat).setNumBaseLearners(2).fit(reg_frame)
        assert torch.isfinite(gbm.predict(reg_frame["features"])).all()


def test_gbm_beats_boosting(clf_frame, clf_frame_test):
    """Reference GBMClassifierSuite.scala:51-87 also asserts GBM beats
    plain AdaBoost on the same budget."""
    import spark_ensemble_amd as sea

    gbm = sea.GBMClassifier().setNumBaseLearners(10).setSeed(3).fit(clf_frame)
    bst = sea.BoostingClassifier().setNumBaseLearners(10).setSeed(3).fit(clf_frame)
    y = clf_frame_test["label"]
    acc_g = float((gbm.transform(clf_frame_test)["prediction"] == y).float().mean())
    acc_b = float((bst.transform(clf_frame_test)["prediction"] == y).float().mean())
    assert acc_g >= acc_b - 0.02, (acc_g, acc_b)
