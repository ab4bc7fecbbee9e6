"""Stacking suites (reference StackingClassifierSuite / StackingRegressorSuite:
ensemble beats its best member; round-trip persistence) plus OOF-specific
checks new to this framework (BASELINE config 5)."""

import pytest
import torch

from spark_ensemble_amd import (
    StackingClassificationModel,
    StackingClassifier,
    StackingRegressionModel,
    StackingRegressor,
)
from spark_ensemble_amd.models import (
    DecisionTreeClassifier,
    DecisionTreeRegressor,
    LinearRegression,
    LogisticRegression,
)


def _acc_pred(pred, frame):
    return float((pred == frame["label"]).float().mean())


def _rmse_model(model, frame):
    return float(((model.predict(frame["features"]) - frame["label"]) ** 2).mean() ** 0.5)


def _stack_reg(in_sample):
    return (
        StackingRegressor()
        .setBaseLearners(
            [DecisionTreeRegressor().setMaxDepth(5), LinearRegression()]
        )
        .setStacker(LinearRegression())
        .setNumFolds(3)
        .setInSample(in_sample)
    )


@pytest.mark.parametrize("in_sample", [False, True])
def test_stacking_regressor_beats_best_member(reg_frame, reg_frame_test, in_sample):
    st = _stack_reg(in_sample).fit(reg_frame)
    member_rmses = [_rmse_model(m, reg_frame_test) for m in st._models]
    assert _rmse_model(st, reg_frame_test) <= min(member_rmses) + 0.05


@pytest.mark.parametrize("method", ["class", "raw", "proba"])
def test_stacking_classifier_methods(clf_frame, clf_frame_test, method):
    st = (
        StackingClassifier()
        .setBaseLearners(
            [DecisionTreeClassifier().setMaxDepth(5), LogisticRegression().setMaxIter(30)]
        )
        .setStacker(LogisticRegression().setMaxIter(30))
        .setStackMethod(method)
        .setNumFolds(3)
        .fit(clf_frame)
    )
    out = st.transform(clf_frame_test)
    assert _acc_pred(out["prediction"], clf_frame_test) > 0.5


def test_oof_close_to_in_sample_on_test(reg_frame, reg_frame_test):
    oof = _stack_reg(False).fit(reg_frame)
    ins = _stack_reg(True).fit(reg_frame)
    assert abs(_rmse_model(oof, reg_frame_test) - _rmse_model(ins, reg_frame_test)) < 0.3


def test_stacking_regressor_roundtrip(tmp_path, reg_frame):
    st = _stack_reg(True).fit(reg_frame)
    p = str(tmp_path / "st")
    st.save(p)
    loaded = StackingRegressionModel.load(p)
    assert torch.allclose(
        st.predict(reg_frame["features"]), loaded.predict(reg_frame["features"]),
        atol=1e-5,
    )


def test_stacking_classifier_roundtrip(tmp_path, clf_frame):
    st = (
        StackingClassifier()
        .setBaseLearners([DecisionTreeClassifier().setMaxDepth(4)])
        .setStacker(LogisticRegression().setMaxIter(20))
        .setStackMethod("proba")
        .setInSample(True)
        .fit(clf_frame)
    )
    p = str(tmp_path / "stc")
    st.save(p)
    loaded = StackingClassificationModel.load(p)
    assert torch.equal(
        st.transform(clf_frame)["prediction"],
        loaded.transform(clf_frame)["prediction"],
    )


def test_stacking_estimator_roundtrip(tmp_path):
    est = _stack_reg(False)
    p = str(tmp_path / "este")
    est.save(p)
    est2 = StackingRegressor.load(p)
    assert len(est2.getBaseLearners()) == 2
    assert est2.getOrDefault("numFolds") == 3
