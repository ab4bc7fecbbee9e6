"""Pipeline / ParamGridBuilder / evaluators / CrossValidator (the Spark ML
workflow the reference's docs drive its estimators through — reference
docs/example.md)."""

import math

import torch

import spark_ensemble_amd as sea
from spark_ensemble_amd.tuning import (
    CrossValidator,
    MulticlassClassificationEvaluator,
    ParamGridBuilder,
    Pipeline,
    RegressionEvaluator,
    TrainValidationSplit,
)
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression


def test_param_grid_builder_cartesian():
    grid = (
        ParamGridBuilder()
        .addGrid("learningRate", [0.1, 0.3])
        .addGrid("numBaseLearners", [2, 3, 4])
        .baseOn(seed=7)
        .build()
    )
    assert len(grid) == 6
    assert all(g["seed"] == 7 for g in grid)
    assert {g["learningRate"] for g in grid} == {0.1, 0.3}


def test_regression_evaluator_metrics():
    df = synthetic_regression(500, 6, seed=2)
    pred = df["label"] + 0.5
    out = df.withColumn("prediction", pred)
    assert abs(RegressionEvaluator("rmse").evaluate(out) - 0.5) < 1e-6
    assert abs(RegressionEvaluator("mae").evaluate(out) - 0.5) < 1e-6
    r2 = RegressionEvaluator("r2").evaluate(out)
    var = float(df["label"].var(unbiased=False))
    assert abs(r2 - (1 - 0.25 / var)) < 1e-5


def test_multiclass_evaluator_accuracy_and_f1():
    y = torch.tensor([0, 0, 1, 1, 2, 2], dtype=torch.float32)
    p = torch.tensor([0, 1, 1, 1, 2, 0], dtype=torch.float32)
    from spark_ensemble_amd.frame import TensorFrame

    df = TensorFrame(label=y, prediction=p,
                     features=torch.zeros(6, 1))
    acc = MulticlassClassificationEvaluator("accuracy").evaluate(df)
    assert abs(acc - 4 / 6) < 1e-9
    f1 = MulticlassClassificationEvaluator("f1").evaluate(df)
    assert 0.0 < f1 < 1.0


def test_pipeline_fit_transform():
    df = synthetic_classification(400, 8, k=2, seed=4)
    pipe = Pipeline(stages=[sea.GBMClassifier().setNumBaseLearners(3)])
    pm = pipe.fit(df)
    out = pm.transform(df)
    assert "prediction" in out
    acc = float((out["prediction"] == df["label"]).float().mean())
    assert acc > 0.6


def test_cross_validator_selects_and_refits():
    df = synthetic_classification(600, 8, k=2, seed=5)
    cv = CrossValidator(
        estimator=sea.GBMClassifier().setSeed(3),
        estimatorParamMaps=ParamGridBuilder()
        .addGrid("numBaseLearners", [1, 5])
        .build(),
        evaluator=MulticlassClassificationEvaluator("accuracy"),
        numFolds=3,
        seed=11,
    )
    m = cv.fit(df)
    assert len(m.avgMetrics) == 2
    assert m.bestIndex == max(range(2), key=lambda i: m.avgMetrics[i])
    out = m.transform(df)
    assert "prediction" in out


def test_train_validation_split():
    df = synthetic_regression(600, 8, seed=6)
    tvs = TrainValidationSplit(
        estimator=sea.GBMRegressor().setSeed(3),
        estimatorParamMaps=ParamGridBuilder()
        .addGrid("numBaseLearners", [1, 4])
        .build(),
        evaluator=RegressionEvaluator("rmse"),
        trainRatio=0.75,
        seed=12,
    )
    m = tvs.fit(df)
    assert len(m.avgMetrics) == 2
    # 4 boosting rounds must beat 1 on held-out rmse for this easy data
    assert m.bestIndex == 1, m.avgMetrics


def test_pipeline_save_load(tmp_path):
    from spark_ensemble_amd.tuning import Pipeline, PipelineModel

    df = synthetic_classification(400, 8, k=2, seed=4)
    pm = Pipeline(stages=[sea.GBMClassifier().setNumBaseLearners(2)]).fit(df)
    p = str(tmp_path / "pipe")
    pm.save(p)
    pm2 = PipelineModel.load(p)
    a = pm.transform(df)["rawPrediction"]
    b = pm2.transform(df)["rawPrediction"]
    assert torch.allclose(a, b, rtol=1e-6, atol=1e-7)

    # estimator pipeline round-trips too
    est = Pipeline(stages=[sea.GBMRegressor().setNumBaseLearners(2)])
    pe = str(tmp_path / "pipe_est")
    est.save(pe)
    est2 = Pipeline.load(pe)
    assert len(est2.getStages()) == 1


def test_cross_validator_model_save_load(tmp_path):
    from spark_ensemble_amd.tuning import CrossValidator, CrossValidatorModel

    df = synthetic_regression(400, 6, seed=8)
    cv = CrossValidator(
        estimator=sea.GBMRegressor().setSeed(1),
        estimatorParamMaps=[{"numBaseLearners": 1}, {"numBaseLearners": 3}],
        evaluator=RegressionEvaluator("rmse"),
        numFolds=2, seed=3,
    )
    m = cv.fit(df)
    p = str(tmp_path / "cv")
    m.save(p)
    m2 = CrossValidatorModel.load(p)
    assert m2.bestIndex == m.bestIndex
    assert m2.avgMetrics == [float(v) for v in m.avgMetrics]
    a = m.transform(df)["prediction"]
    b = m2.transform(df)["prediction"]
    assert torch.allclose(a, b, rtol=1e-6, atol=1e-7)


def test_evaluators_match_sklearn_exactly():
    """Metric formulas pinned against scikit-learn (the reference pins
    its evaluators to Spark MLlib's — SURVEY §4.1 cross-library genre)."""
    import numpy as np
    from sklearn.metrics import (
        accuracy_score, f1_score, mean_absolute_error, mean_squared_error,
        precision_score, r2_score, recall_score,
    )

    from spark_ensemble_amd.frame import TensorFrame
    from spark_ensemble_amd.tuning import (
        MulticlassClassificationEvaluator, RegressionEvaluator,
    )

    g = torch.Generator().manual_seed(11)
    y = torch.randint(0, 4, (2000,), generator=g).float()
    p = torch.where(torch.rand(2000, generator=g) < 0.7, y,
                    torch.randint(0, 4, (2000,), generator=g).float())
    df = TensorFrame(label=y, prediction=p)
    yn, pn = y.numpy(), p.numpy()
    assert abs(MulticlassClassificationEvaluator("accuracy").evaluate(df)
               - accuracy_score(yn, pn)) < 1e-6
    assert abs(MulticlassClassificationEvaluator("f1").evaluate(df)
               - f1_score(yn, pn, average="weighted")) < 1e-5
    assert abs(MulticlassClassificationEvaluator("weightedPrecision").evaluate(df)
               - precision_score(yn, pn, average="weighted")) < 1e-5
    assert abs(MulticlassClassificationEvaluator("weightedRecall").evaluate(df)
               - recall_score(yn, pn, average="weighted")) < 1e-5

    yr = torch.randn(2000, generator=g)
    pr = yr + 0.3 * torch.randn(2000, generator=g)
    dfr = TensorFrame(label=yr, prediction=pr)
    yrn, prn = yr.numpy(), pr.numpy()
    assert abs(RegressionEvaluator("rmse").evaluate(dfr)
               - mean_squared_error(yrn, prn) ** 0.5) < 1e-5
    assert abs(RegressionEvaluator("mae").evaluate(dfr)
               - mean_absolute_error(yrn, prn)) < 1e-5
    assert abs(RegressionEvaluator("r2").evaluate(dfr)
               - r2_score(yrn, prn)) < 1e-5
