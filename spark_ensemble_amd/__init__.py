"""spark_ensemble_amd — MI355X-native ensemble learning framework.

A from-scratch rebuild of the capabilities of pierrenodet/spark-ensemble
(scikit-learn-style meta-estimators for Bagging / Boosting / GBM / Stacking)
designed for AMD Instinct MI355X (gfx950): PyTorch-ROCm columnar tensor
frames resident in HBM3E, hand-written CDNA4 HIP kernels for the hot path
(LDS-staged histogram tree building, fused gradient/hessian, batched forest
inference) and RCCL over xGMI for multi-GPU scaling (one process per GPU).

Public API mirrors the reference: ``Estimator.fit(frame) -> Model``,
``Model.transform(frame)``, ``model.save(path)`` / ``Model.load(path)``.
"""

from .frame import TensorFrame  # noqa: F401
from .estimator import Estimator, Model  # noqa: F401

__version__ = "0.1.0"


def __getattr__(name):
    # lazy imports of the public estimator/model surface
    _public = {
        # base learners
        "DecisionTreeClassifier": "spark_ensemble_amd.models",
        "DecisionTreeRegressor": "spark_ensemble_amd.models",
        "DecisionTreeClassificationModel": "spark_ensemble_amd.models",
        "DecisionTreeRegressionModel": "spark_ensemble_amd.models",
        "LinearRegression": "spark_ensemble_amd.models",
        "LinearRegressionModel": "spark_ensemble_amd.models",
        "LogisticRegression": "spark_ensemble_amd.models",
        "LogisticRegressionModel": "spark_ensemble_amd.models",
        "DummyClassifier": "spark_ensemble_amd.models",
        "DummyRegressor": "spark_ensemble_amd.models",
        "DummyClassificationModel": "spark_ensemble_amd.models",
        "DummyRegressionModel": "spark_ensemble_amd.models",
        # meta-estimators: classification
        "BaggingClassifier": "spark_ensemble_amd.classification",
        "BaggingClassificationModel": "spark_ensemble_amd.classification",
        "BoostingClassifier": "spark_ensemble_amd.classification",
        "BoostingClassificationModel": "spark_ensemble_amd.classification",
        "GBMClassifier": "spark_ensemble_amd.classification",
        "GBMClassificationModel": "spark_ensemble_amd.classification",
        "StackingClassifier": "spark_ensemble_amd.classification",
        "StackingClassificationModel": "spark_ensemble_amd.classification",
        # meta-estimators: regression
        "BaggingRegressor": "spark_ensemble_amd.regression",
        "BaggingRegressionModel": "spark_ensemble_amd.regression",
        "BoostingRegressor": "spark_ensemble_amd.regression",
        "BoostingRegressionModel": "spark_ensemble_amd.regression",
        "GBMRegressor": "spark_ensemble_amd.regression",
        "GBMRegressionModel": "spark_ensemble_amd.regression",
        "StackingRegressor": "spark_ensemble_amd.regression",
        "StackingRegressionModel": "spark_ensemble_amd.regression",
    }
    if name in _public:
        import importlib

        mod = importlib.import_module(_public[name])
        return getattr(mod, name)
    raise AttributeError(f"module 'spark_ensemble_amd' has no attribute {name!r}")
