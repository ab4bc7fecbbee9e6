"""Round-state checkpoint / resume (SURVEY.md §5.4 rebuild of the
reference's PeriodicRDDCheckpointer lineage management — here fit() can
actually resume a half-trained GBM from the dump)."""

import os

import pytest
import torch

import spark_ensemble_amd as sea
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression


def _fit_gbm_reg(tmp, k, interval=2, ckpt=True):
    df = synthetic_regression(600, 10, seed=11)
    est = (
        sea.GBMRegressor()
        .setNumBaseLearners(k)
        .setSeed(5)
    )
    if ckpt:
        est.setCheckpointInterval(interval).setCheckpointDir(str(tmp / "ck"))
    return est, df, est.fit(df)


def test_gbm_regressor_resume_matches_straight_fit(tmp_path):
    # full fit, no checkpointing — the golden result
    est0, df, m_full = _fit_gbm_reg(tmp_path, 6, ckpt=False)

    # interrupted fit: run 4 rounds with dumps every 2, then "crash"
    est1, _, _ = _fit_gbm_reg(tmp_path, 4, interval=2)
    assert os.path.exists(tmp_path / "ck" / "state.json")

    # resume: same params, full 6 rounds — must pick up at round 4
    est2 = (
        sea.GBMRegressor()
        .setNumBaseLearners(6)
        .setSeed(5)
        .setCheckpointInterval(2)
        .setCheckpointDir(str(tmp_path / "ck"))
    )
    m_res = est2.fit(df)
    assert est2._instr.history, "resume should still train rounds 4..6"
    assert m_res.numModels == 6

    out_full = m_full.transform(df)["prediction"]
    out_res = m_res.transform(df)["prediction"]
    assert torch.allclose(out_full, out_res, rtol=1e-4, atol=1e-5)


def test_gbm_classifier_resume_matches_straight_fit(tmp_path):
    df = synthetic_classification(600, 10, k=3, seed=3)

    def mk(k, ck):
        e = sea.GBMClassifier().setNumBaseLearners(k).setSeed(7)
        if ck:
            e.setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "ckc"))
        return e

    m_full = mk(4, False).fit(df)
    mk(2, True).fit(df)  # interrupted after 2 rounds (dump at 2)
    m_res = mk(4, True).fit(df)
    a = m_full.transform(df)["probability"]
    b = m_res.transform(df)["probability"]
    assert torch.allclose(a, b, rtol=1e-4, atol=1e-5)


def test_save_load_round_state_roundtrip(tmp_path):
    from spark_ensemble_amd.utils import checkpoint as ck

    df = synthetic_regression(200, 6, seed=1)
    m1 = sea.DummyRegressor().setStrategy("mean").fit(df)
    m2 = sea.DummyRegressor().setStrategy("median").fit(df)
    ck.save_round_state(str(tmp_path / "s"), 2, [m1, m2], [0.5, 0.25],
                        extra={"best_err": 1.5, "v": 1})
    r, models, weights, extra = ck.load_round_state(str(tmp_path / "s"))
    assert r == 2 and weights == [0.5, 0.25]
    assert extra == {"best_err": 1.5, "v": 1}
    x = df["features"]
    assert torch.allclose(models[0].predict(x), m1.predict(x))
    ck.clear(str(tmp_path / "s"))
    assert ck.load_round_state(str(tmp_path / "s")) is None


def test_boosting_resume_matches_straight_fit(tmp_path):
    df = synthetic_classification(600, 10, k=2, seed=13)

    def mk(k, ck):
        e = sea.BoostingClassifier().setNumBaseLearners(k).setSeed(2)
        if ck:
            e.setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "bk"))
        return e

    m_full = mk(5, False).fit(df)
    mk(2, True).fit(df)  # interrupted (dump at round 2)
    m_res = mk(5, True).fit(df)
    a = m_full.transform(df)["rawPrediction"]
    b = m_res.transform(df)["rawPrediction"]
    assert torch.allclose(a, b, rtol=1e-5, atol=1e-6)
