"""Bagging statistical suites (reference BaggingClassifierSuite /
BaggingRegressorSuite: ensemble beats base learner and best member,
diversity among bagged models, round-trip persistence)."""

import itertools

import torch

from spark_ensemble_amd import (
    BaggingClassificationModel,
    BaggingClassifier,
    BaggingRegressionModel,
    BaggingRegressor,
)
from spark_ensemble_amd.ensemble.utils import slice_features
from spark_ensemble_amd.models import DecisionTreeClassifier, DecisionTreeRegressor


def _acc(model, frame):
    out = model.transform(frame)
    return float((out["prediction"] == frame["label"]).float().mean())


def _rmse(model, frame):
    return float(((model.predict(frame["features"]) - frame["label"]) ** 2).mean() ** 0.5)


def test_bagging_classifier_beats_base_learner(clf_frame, clf_frame_test):
    base = DecisionTreeClassifier().setMaxDepth(5)
    single = base.fit(clf_frame)
    bag = (
        BaggingClassifier()
        .setBaseLearner(base)
        .setNumBaseLearners(10)
        .setSubsampleRatio(0.8)
        .setSubspaceRatio(0.8)
        .setVotingStrategy("soft")
        .fit(clf_frame)
    )
    assert _acc(bag, clf_frame_test) >= _acc(single, clf_frame_test) - 0.005


def test_bagging_classifier_beats_best_member(clf_frame, clf_frame_test):
    bag = (
        BaggingClassifier()
        .setBaseLearner(DecisionTreeClassifier().setMaxDepth(5))
        .setNumBaseLearners(10)
        .setSubsampleRatio(0.7)
        .setSubspaceRatio(0.7)
        .setVotingStrategy("soft")
        .fit(clf_frame)
    )
    x = clf_frame_test["features"]
    y = clf_frame_test["label"]
    member_accs = []
    for sub, m in zip(bag._subspaces, bag._models):
        pred = m.transform(
            clf_frame_test.withColumn("features", slice_features(x, sub))
        )["prediction"]
        member_accs.append(float((pred == y).float().mean()))
    assert _acc(bag, clf_frame_test) >= max(member_accs) - 0.02


def test_bagged_models_are_diverse(clf_frame):
    bag = (
        BaggingClassifier()
        .setBaseLearner(DecisionTreeClassifier().setMaxDepth(5))
        .setNumBaseLearners(6)
        .setSubsampleRatio(0.6)
        .setSubspaceRatio(0.6)
        .fit(clf_frame)
    )
    x = clf_frame["features"]
    preds = [
        m.transform(clf_frame.withColumn("features", slice_features(x, sub)))[
            "prediction"
        ]
        for sub, m in zip(bag._subspaces, bag._models)
    ]
    agreements = [
        float((a == b).float().mean()) for a, b in itertools.combinations(preds, 2)
    ]
    # reference asserts pairwise agreement < 0.85 (BaggingClassifierSuite:114-155)
    assert sum(agreements) / len(agreements) < 0.9


def test_bagging_regressor_beats_base(reg_frame, reg_frame_test):
    base = DecisionTreeRegressor().setMaxDepth(5)
    single = base.fit(reg_frame)
    bag = (
        BaggingRegressor()
        .setBaseLearner(base)
        .setNumBaseLearners(10)
        .setSubsampleRatio(0.8)
        .setSubspaceRatio(0.9)
        .fit(reg_frame)
    )
    assert _rmse(bag, reg_frame_test) <= _rmse(single, reg_frame_test) + 1e-6


def test_hard_vs_soft_voting_close(bin_frame, bin_frame_test):
    common = dict(numBaseLearners=8, subsampleRatio=0.8)
    hard = (
        BaggingClassifier()
        .setBaseLearner(DecisionTreeClassifier().setMaxDepth(5))
        .setVotingStrategy("hard")
        .copy(common)
        .fit(bin_frame)
    )
    soft = (
        BaggingClassifier()
        .setBaseLearner(DecisionTreeClassifier().setMaxDepth(5))
        .setVotingStrategy("soft")
        .copy(common)
        .fit(bin_frame)
    )
    assert abs(_acc(hard, bin_frame_test) - _acc(soft, bin_frame_test)) < 0.05


def test_bagging_classifier_roundtrip(tmp_path, clf_frame):
    bag = (
        BaggingClassifier()
        .setBaseLearner(DecisionTreeClassifier().setMaxDepth(4))
        .setNumBaseLearners(3)
        .fit(clf_frame)
    )
    p = str(tmp_path / "bag")
    bag.write.overwrite().save(p)
    loaded = BaggingClassificationModel.load(p)
    o1 = bag.transform(clf_frame)
    o2 = loaded.transform(clf_frame)
    assert torch.equal(o1["prediction"], o2["prediction"])
    assert torch.allclose(o1["rawPrediction"], o2["rawPrediction"])


def test_bagging_regressor_roundtrip(tmp_path, reg_frame):
    bag = (
        BaggingRegressor()
        .setBaseLearner(DecisionTreeRegressor().setMaxDepth(4))
        .setNumBaseLearners(3)
        .setSubspaceRatio(0.7)
        .fit(reg_frame)
    )
    p = str(tmp_path / "bagr")
    bag.save(p)
    loaded = BaggingRegressionModel.load(p)
    assert torch.allclose(
        bag.predict(reg_frame["features"]), loaded.predict(reg_frame["features"])
    )


def test_bagging_estimator_roundtrip(tmp_path):
    est = (
        BaggingRegressor()
        .setBaseLearner(DecisionTreeRegressor().setMaxDepth(7))
        .setNumBaseLearners(4)
        .setSubsampleRatio(0.5)
    )
    p = str(tmp_path / "est")
    est.save(p)
    est2 = BaggingRegressor.load(p)
    assert est2.getNumBaseLearners() == 4
    assert est2.getSubsampleRatio() == 0.5
    assert est2.getBaseLearner().getOrDefault("maxDepth") == 7
