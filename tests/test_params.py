"""Param system unit tests (config surface of SURVEY.md section 5.6)."""

import pytest

from spark_ensemble_amd import BaggingClassifier, GBMRegressor
from spark_ensemble_amd.models import DecisionTreeClassifier
from spark_ensemble_amd.params import ParamValidators


def test_defaults_match_reference():
    # reference BaggingParams.scala:27-36, GBMParams.scala:121-129
    b = BaggingClassifier()
    assert b.getNumBaseLearners() == 10
    assert b.getReplacement() is True
    assert b.getSubsampleRatio() == 1.0
    assert b.getSubspaceRatio() == 1.0
    assert b.getVotingStrategy() == "hard"
    g = GBMRegressor()
    assert g.getOrDefault("optimizedWeights") is True
    assert g.getOrDefault("updates") == "gradient"
    assert g.getOrDefault("learningRate") == 1.0
    assert g.getOrDefault("maxIter") == 100
    assert g.getOrDefault("numRounds") == 1
    assert g.getOrDefault("validationTol") == 0.01
    assert g.getOrDefault("replacement") is False  # GBM overrides to False
    assert g.getLoss() == "squared"


def test_validators():
    g = GBMRegressor()
    with pytest.raises(ValueError):
        g.set("learningRate", -1.0)
    with pytest.raises(ValueError):
        g.set("loss", "nonsense")
    with pytest.raises(ValueError):
        g.set("subsampleRatio", 0.0)
    g.set("subsampleRatio", 0.5)
    assert g.getSubsampleRatio() == 0.5


def test_string_params_lowercased():
    g = GBMRegressor().setLoss("SQUARED")
    assert g.getLoss() == "squared"


def test_copy_with_extra_and_nested():
    dt = DecisionTreeClassifier().setMaxDepth(7)
    b = BaggingClassifier().setBaseLearner(dt).setNumBaseLearners(5)
    c = b.copy({"numBaseLearners": 3})
    assert c.getNumBaseLearners() == 3
    assert b.getNumBaseLearners() == 5
    # nested estimator must be deep-copied (reference BaggingRegressor.scala:111-115)
    assert c.getBaseLearner() is not dt
    assert c.getBaseLearner().getOrDefault("maxDepth") == 7


def test_explain_params():
    text = GBMRegressor().explainParams()
    assert "learningRate" in text and "default" in text


def test_unknown_param_raises():
    with pytest.raises(AttributeError):
        GBMRegressor().set("bogus", 1)


def test_fit_with_param_map_list():
    import spark_ensemble_amd as sea
    from spark_ensemble_amd.utils.io import synthetic_regression

    df = synthetic_regression(300, 6, seed=2)
    models = sea.GBMRegressor().fit(
        df, [{"numBaseLearners": 1}, {"numBaseLearners": 3}]
    )
    assert len(models) == 2
    assert models[0].numModels == 1 and models[1].numModels == 3


def test_mllib_metadata_layout(tmp_path):
    """Pin the on-disk contract (reference §3.4 layout): one JSON line in
    metadata/part-00000 with class/uid/paramMap, nested model-$i dirs."""
    import json
    import os

    import spark_ensemble_amd as sea
    from spark_ensemble_amd.utils.io import synthetic_regression

    df = synthetic_regression(200, 5, seed=1)
    m = sea.BaggingRegressor().setNumBaseLearners(2).fit(df)
    p = tmp_path / "m"
    m.save(str(p))

    meta_file = p / "metadata" / "part-00000"
    assert meta_file.exists()
    lines = meta_file.read_text().strip().split("\n")
    assert len(lines) == 1
    meta = json.loads(lines[0])
    assert meta["class"].endswith("BaggingRegressionModel")
    assert "uid" in meta and "paramMap" in meta
    assert meta["numModels"] == 2
    assert (p / "model-0" / "metadata" / "part-00000").exists()
    assert (p / "model-1").is_dir()
    assert (p / "data-0" / "part-00000").exists()
    row = json.loads((p / "data-0" / "part-00000").read_text().strip())
    assert "subspace" in row


def test_model_public_accessors():
    """Every ensemble model exposes the reference's public fields
    (.models, .weights / .subspaces / .stack)."""
    import torch
    from spark_ensemble_amd import (
        BaggingRegressor, BoostingRegressor, GBMClassifier, GBMRegressor,
        StackingRegressor,
    )
    from spark_ensemble_amd.frame import TensorFrame
    from spark_ensemble_amd.models import LinearRegression

    g = torch.Generator().manual_seed(0)
    x = torch.randn(400, 5, generator=g)
    y = x[:, 0] - 0.5 * x[:, 1]
    df = TensorFrame(features=x, label=y)

    m = GBMRegressor().setNumBaseLearners(2).fit(df)
    assert len(m.models) == 2 and len(m.weights) == 2
    b = BaggingRegressor().setNumBaseLearners(2).fit(df)
    assert len(b.models) == 2 and len(b.subspaces) == 2
    bo = BoostingRegressor().setNumBaseLearners(2).fit(df)
    assert len(bo.models) == len(bo.weights) > 0
    st = (
        StackingRegressor()
        .setBaseLearners([LinearRegression()])
        .setStacker(LinearRegression())
        .setNumFolds(2)
        .fit(df)
    )
    assert len(st.models) == 1 and st.stack is not None

    yc = (y > 0).float()
    dfc = TensorFrame(features=x, label=yc)
    c = GBMClassifier().setNumBaseLearners(2).fit(dfc)
    assert len(c.models) == 2 and isinstance(c.models[0], list)
    assert len(c.weights) == 2
