"""Estimator / Model class hierarchy.

Mirrors the Spark ML contract the reference's meta-estimators are written
against (reference: the ``Predictor``/``PredictionModel``/
``ProbabilisticClassifier`` hierarchy referenced throughout
ensemble/package.scala:32-65): ``Estimator.fit(frame) -> Model``,
``Model.transform(frame)`` appends prediction columns, ``save``/``load``
round-trips through the MLlib directory layout (persistence.py).

Compute happens in the subclasses through the ops layer (torch reference on
CPU, HIP kernels on gfx950).
"""

from __future__ import annotations

import os
from abc import abstractmethod
from typing import Optional

import torch

from . import persistence
from .frame import TensorFrame
from .params import Params


class Identifiable(Params):
    pass


class PipelineStage(Identifiable):
    # -- persistence: shared writer plumbing ------------------------------
    def save(self, path: str, overwrite: bool = False):
        if os.path.exists(path):
            if not overwrite:
                raise FileExistsError(
                    f"Path {path} already exists; use overwrite=True"
                )
        os.makedirs(path, exist_ok=True)
        self._save_impl(path)

    @property
    def write(self):
        return _Writer(self)

    def _save_impl(self, path: str):
        persistence.save_metadata(self, path)

    @classmethod
    def load(cls, path: str):
        inst = persistence.load_instance(path)
        if not isinstance(inst, cls):
            raise TypeError(f"loaded {type(inst).__name__}, expected {cls.__name__}")
        return inst

    @classmethod
    def read(cls):
        return _Reader(cls)

    @classmethod
    def _load_from(cls, path: str, meta: dict):
        inst = cls()
        inst.uid = meta["uid"]
        inst._setFromJson(meta.get("paramMap", {}))
        inst._load_extra(path, meta)
        return inst

    def _load_extra(self, path: str, meta: dict):
        pass


class _Writer:
    """``model.write.overwrite().save(path)`` fluent API (Spark MLWriter)."""

    def __init__(self, instance):
        self._instance = instance
        self._overwrite = False

    def overwrite(self):
        self._overwrite = True
        return self

    def save(self, path: str):
        self._instance.save(path, overwrite=self._overwrite)


class _Reader:
    def __init__(self, cls):
        self._cls = cls

    def load(self, path: str):
        return self._cls.load(path)


class Estimator(PipelineStage):
    def fit(self, dataset: TensorFrame, params: Optional[dict] = None):
        """Fit on ``dataset``. ``params`` may be a dict of overrides (one
        model returned) or a LIST of dicts (one model per map — the Spark
        ``fit(dataset, paramMaps)`` overload)."""
        if isinstance(params, (list, tuple)):
            return [self.fit(dataset, pm) for pm in params]
        est = self.copy(params) if params else self
        return est._fit(dataset)

    @abstractmethod
    def _fit(self, dataset: TensorFrame) -> "Model":
        ...


class Transformer(PipelineStage):
    @abstractmethod
    def transform(self, dataset: TensorFrame) -> TensorFrame:
        ...


class Model(Transformer):
    pass


# ---------------------------------------------------------------------------
# Predictor layer (featuresCol / labelCol / predictionCol / weightCol)
# ---------------------------------------------------------------------------


class _PredictorParams(Params):
    def _declare_params(self):
        super()._declare_params()
        self.featuresCol = self._str_param(
            "featuresCol", "features column name", lower=False
        )
        self.labelCol = self._str_param("labelCol", "label column name", lower=False)
        self.predictionCol = self._str_param(
            "predictionCol", "prediction column name", lower=False
        )
        self.weightCol = self._str_param(
            "weightCol", "instance weight column name", lower=False
        )
        self._setDefault(
            featuresCol="features", labelCol="label", predictionCol="prediction"
        )

    # accessors (Spark-style names)
    def getFeaturesCol(self):
        return self.getOrDefault("featuresCol")

    def getLabelCol(self):
        return self.getOrDefault("labelCol")

    def getPredictionCol(self):
        return self.getOrDefault("predictionCol")

    def getWeightCol(self):
        return self.getOrNone("weightCol")

    def setFeaturesCol(self, v):
        return self.set("featuresCol", v)

    def setLabelCol(self, v):
        return self.set("labelCol", v)

    def setPredictionCol(self, v):
        return self.set("predictionCol", v)

    def setWeightCol(self, v):
        return self.set("weightCol", v)

    # -- instance extraction ---------------------------------------------
    def _extract_xyw(self, dataset: TensorFrame, require_label: bool = True):
        """(features [N,F] f32, label [N] f32, weight [N] f32) — the analog
        of Spark's ``extractInstances`` (used at reference
        BaggingRegressor.scala:136)."""
        x = dataset[self.getFeaturesCol()].float()
        label_col = self.getLabelCol()
        if label_col in dataset:
            y = dataset[label_col].float()
        elif require_label:
            raise KeyError(f"label column {label_col!r} not found in dataset")
        else:
            y = None
        wcol = self.getWeightCol()
        if wcol and wcol in dataset:
            w = dataset[wcol].float()
        else:
            w = torch.ones(x.shape[0], dtype=torch.float32, device=x.device)
        return x, y, w


class Predictor(Estimator, _PredictorParams):
    """Supervised estimator over a vector features column."""

    def supports_weight(self) -> bool:
        """Whether this learner honors instance weights (the reference checks
        base learners for a weightCol param — StackingRegressor.scala:112-119)."""
        return True


class PredictionModel(Model, _PredictorParams):
    @abstractmethod
    def predict(self, features: torch.Tensor) -> torch.Tensor:
        """Batched prediction: [N, F] -> [N]."""
        ...

    @property
    def numFeatures(self) -> int:
        return getattr(self, "_num_features", -1)

    def _check_features(self, x: torch.Tensor):
        nf = self.numFeatures
        if nf > 0 and x.shape[1] != nf:
            raise ValueError(
                f"{type(self).__name__} was trained on {nf} features but "
                f"received {x.shape[1]}"
            )

    def transform(self, dataset: TensorFrame) -> TensorFrame:
        x = dataset[self.getFeaturesCol()].float()
        self._check_features(x)
        return dataset.withColumn(self.getPredictionCol(), self.predict(x))


# ---------------------------------------------------------------------------
# Classification layer
# ---------------------------------------------------------------------------


class _ClassifierParams(_PredictorParams):
    def _declare_params(self):
        super()._declare_params()
        self.rawPredictionCol = self._str_param(
            "rawPredictionCol", "raw prediction (margin) column name", lower=False
        )
        self._setDefault(rawPredictionCol="rawPrediction")

    def getRawPredictionCol(self):
        return self.getOrDefault("rawPredictionCol")

    def setRawPredictionCol(self, v):
        return self.set("rawPredictionCol", v)


class _ProbabilisticClassifierParams(_ClassifierParams):
    def _declare_params(self):
        super()._declare_params()
        self.probabilityCol = self._str_param(
            "probabilityCol", "class probability column name", lower=False
        )
        self._setDefault(probabilityCol="probability")

    def getProbabilityCol(self):
        return self.getOrDefault("probabilityCol")

    def setProbabilityCol(self, v):
        return self.set("probabilityCol", v)


class Classifier(Predictor, _ClassifierParams):
    def _get_num_classes(self, dataset: TensorFrame) -> int:
        y = dataset[self.getLabelCol()]
        return int(y.max().item()) + 1


class ProbabilisticClassifier(Classifier, _ProbabilisticClassifierParams):
    pass


class ClassificationModel(PredictionModel, _ClassifierParams):
    @property
    def numClasses(self) -> int:
        return getattr(self, "_num_classes", -1)

    @abstractmethod
    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        """[N, F] -> [N, K] raw margins."""
        ...

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        return self.predictRaw(features).argmax(dim=1).float()

    def transform(self, dataset: TensorFrame) -> TensorFrame:
        x = dataset[self.getFeaturesCol()].float()
        self._check_features(x)
        raw = self.predictRaw(x)
        out = dataset.withColumn(self.getRawPredictionCol(), raw)
        out = out.withColumn(self.getPredictionCol(), raw.argmax(dim=1).float())
        return out


class ProbabilisticClassificationModel(
    ClassificationModel, _ProbabilisticClassifierParams
):
    @abstractmethod
    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        ...

    def predictProbability(self, features: torch.Tensor) -> torch.Tensor:
        return self.raw2probabilityInPlace(self.predictRaw(features).clone())

    def transform(self, dataset: TensorFrame) -> TensorFrame:
        x = dataset[self.getFeaturesCol()].float()
        self._check_features(x)
        raw = self.predictRaw(x)
        prob = self.raw2probabilityInPlace(raw.clone())
        out = dataset.withColumn(self.getRawPredictionCol(), raw)
        out = out.withColumn(self.getProbabilityCol(), prob)
        out = out.withColumn(self.getPredictionCol(), prob.argmax(dim=1).float())
        return out


class RegressionModel(PredictionModel):
    pass


class Regressor(Predictor):
    pass
