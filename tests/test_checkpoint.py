"""Round-state checkpoint / resume (SURVEY.md §5.4 rebuild of the
reference's PeriodicRDDCheckpointer lineage management — here fit() can
actually resume a half-trained GBM from the dump).

Resume validity contract: a checkpoint carries a fingerprint
of the estimator params + dataset; fit() only resumes on a match, and a
COMPLETED fit clears its own checkpoint dir — so checkpointInterval can
never change a fitted result, matching the reference's semantics where
checkpointing is transparent (PeriodicRDDCheckpointer is lineage-only).
Crashes are simulated by raising from the instrumentation hook mid-fit.
"""

import os

import pytest
import torch

import spark_ensemble_amd as sea
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression


class _InjectedCrash(Exception):
    pass


def _fit_until_crash(est, df, crash_round, monkeypatch):
    """Run est.fit(df) but raise after instrumentation logs round
    ``crash_round`` (0-based) — i.e. after that round finished and any
    due checkpoint dumps for earlier rounds were written."""
    from spark_ensemble_amd.utils.instrumentation import Instrumentation

    orig = Instrumentation.log_round

    def patched(self, i, **kw):
        orig(self, i, **kw)
        if i >= crash_round:
            raise _InjectedCrash()

    monkeypatch.setattr(Instrumentation, "log_round", patched)
    with pytest.raises(_InjectedCrash):
        est.fit(df)
    monkeypatch.undo()


def test_gbm_regressor_resume_matches_straight_fit(tmp_path, monkeypatch):
    df = synthetic_regression(600, 10, seed=11)

    def mk(ck):
        e = sea.GBMRegressor().setNumBaseLearners(6).setSeed(5)
        if ck:
            e.setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "ck"))
        return e

    m_full = mk(False).fit(df)  # golden straight fit

    # interrupted fit: crash after round 5 (dump exists for rounds 1..4)
    _fit_until_crash(mk(True), df, crash_round=4, monkeypatch=monkeypatch)
    assert os.path.exists(tmp_path / "ck" / "state.json")

    # resume: same params — must pick up at round 4 and finish
    est2 = mk(True)
    m_res = est2.fit(df)
    assert est2._instr.history, "resume should still train rounds 4..6"
    assert m_res.numModels == 6
    # a completed fit clears its resume state
    assert not os.path.exists(tmp_path / "ck" / "state.json")

    out_full = m_full.transform(df)["prediction"]
    out_res = m_res.transform(df)["prediction"]
    assert torch.allclose(out_full, out_res, rtol=1e-4, atol=1e-5)


def test_gbm_classifier_resume_matches_straight_fit(tmp_path, monkeypatch):
    df = synthetic_classification(600, 10, k=3, seed=3)

    def mk(ck):
        e = sea.GBMClassifier().setNumBaseLearners(4).setSeed(7)
        if ck:
            e.setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "ckc"))
        return e

    m_full = mk(False).fit(df)
    _fit_until_crash(mk(True), df, crash_round=2, monkeypatch=monkeypatch)
    m_res = mk(True).fit(df)
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


def test_fingerprint_mismatch_ignores_checkpoint(tmp_path):
    from spark_ensemble_amd.utils import checkpoint as ck

    df = synthetic_regression(200, 6, seed=1)
    m1 = sea.DummyRegressor().setStrategy("mean").fit(df)
    ck.save_round_state(str(tmp_path / "f"), 1, [m1], [1.0], fingerprint="aaa")
    assert ck.load_round_state(str(tmp_path / "f"), "aaa") is not None
    assert ck.load_round_state(str(tmp_path / "f"), "bbb") is None
    # no-fingerprint callers (legacy/manual) still load
    assert ck.load_round_state(str(tmp_path / "f")) is not None


def test_unlisted_model_dirs_not_adopted(tmp_path):
    """A stale model dir not listed in state.json must not be loaded."""
    from spark_ensemble_amd.utils import checkpoint as ck

    df = synthetic_regression(200, 6, seed=1)
    m1 = sea.DummyRegressor().setStrategy("mean").fit(df)
    ck.save_round_state(str(tmp_path / "u"), 1, [m1], [1.0])
    # plant a stale partial dir that a pre-fix loader would have walked into
    os.makedirs(tmp_path / "u" / "model-1")
    r, models, _, _ = ck.load_round_state(str(tmp_path / "u"))
    assert r == 1 and len(models) == 1


def test_changed_params_never_resume_stale_state(tmp_path, monkeypatch):
    """Crash a fit, then run a fit with DIFFERENT params on the same
    checkpointDir: it must ignore the stale dump and equal a fresh fit
    (stale resume silently returned a wrong model)."""
    df = synthetic_regression(600, 10, seed=11)

    crash_est = (
        sea.GBMRegressor().setNumBaseLearners(6).setSeed(5)
        .setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "cp"))
    )
    _fit_until_crash(crash_est, df, crash_round=4, monkeypatch=monkeypatch)
    assert os.path.exists(tmp_path / "cp" / "state.json")

    # different seed → different sampling → stale models are wrong for it
    est2 = (
        sea.GBMRegressor().setNumBaseLearners(6).setSeed(99)
        .setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "cp"))
    )
    m2 = est2.fit(df)
    m_fresh = sea.GBMRegressor().setNumBaseLearners(6).setSeed(99).fit(df)
    a = m2.transform(df)["prediction"]
    b = m_fresh.transform(df)["prediction"]
    assert torch.allclose(a, b, rtol=1e-4, atol=1e-5)


def test_completed_fit_then_longer_refit_is_fresh(tmp_path):
    """The exact ADVICE scenario: complete a k=4 fit with a checkpoint dir,
    then fit k=6 on the same dir — the second fit must NOT skip rounds."""
    df = synthetic_regression(600, 10, seed=11)

    def mk(k, ck):
        e = sea.GBMRegressor().setNumBaseLearners(k).setSeed(5)
        if ck:
            e.setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "cc"))
        return e

    mk(4, True).fit(df)  # completes → clears its state
    assert not os.path.exists(tmp_path / "cc" / "state.json")
    m6 = mk(6, True).fit(df)
    m6_fresh = mk(6, False).fit(df)
    a = m6.transform(df)["prediction"]
    b = m6_fresh.transform(df)["prediction"]
    assert torch.allclose(a, b, rtol=1e-4, atol=1e-5)


def test_boosting_resume_matches_straight_fit(tmp_path, monkeypatch):
    df = synthetic_classification(600, 10, k=2, seed=13)

    def mk(ck):
        e = sea.BoostingClassifier().setNumBaseLearners(5).setSeed(2)
        if ck:
            e.setCheckpointInterval(2).setCheckpointDir(str(tmp_path / "bk"))
        return e

    m_full = mk(False).fit(df)
    _fit_until_crash(mk(True), df, crash_round=3, monkeypatch=monkeypatch)
    m_res = mk(True).fit(df)
    a = m_full.transform(df)["rawPrediction"]
    b = m_res.transform(df)["rawPrediction"]
    assert torch.allclose(a, b, rtol=1e-5, atol=1e-6)
