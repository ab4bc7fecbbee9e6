"""Instrumentation subsystem (SURVEY.md §5.1 rebuild): every sequential
meta-estimator records a structured per-round history (round, loss/weight,
ms) on the estimator during fit — the analog of the reference's Spark
``Instrumentation`` ``logNamedValue`` calls (BoostingClassifier.scala:182)."""

import logging

import spark_ensemble_amd as sea
from spark_ensemble_amd.utils.instrumentation import Instrumentation, logger
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression


def test_gbm_round_history():
    df = synthetic_regression(400, 8, seed=3)
    est = sea.GBMRegressor().setNumBaseLearners(4)
    est._fit(df)
    hist = est._instr.history
    assert len(hist) == 4
    assert all("weight" in r and "ms" in r and r["round"] == i
               for i, r in enumerate(hist))
    assert hist[-1]["ms"] >= hist[0]["ms"]


def test_boosting_round_history_and_log_emission(caplog):
    df = synthetic_classification(400, 8, k=2, seed=5)
    est = sea.BoostingClassifier().setNumBaseLearners(3)
    with caplog.at_level(logging.INFO, logger="spark_ensemble_amd"):
        est._fit(df)
    hist = est._instr.history
    assert 1 <= len(hist) <= 3
    assert all("error" in r and "sum_w" in r for r in hist)
    assert any("round" in rec.message for rec in caplog.records)


def test_timed_accumulates():
    class E:
        uid = "e"

    instr = Instrumentation(E())
    with instr.timed("phase"):
        sum(range(1000))
    with instr.timed("phase"):
        sum(range(1000))
    assert instr.timers["phase"] > 0.0
    assert set(instr.timers) == {"phase"}
