"""Baseline (Dummy) models — GBM init models and sanity baselines.

Re-implements reference regression/DummyRegressor.scala (strategies
{mean, median, quantile, constant}; train computes the statistic, the model
predicts the constant) and classification/DummyClassifier.scala (strategies
{uniform, prior, constant}; prior = class-frequency log-priors).  The single
Spark SQL aggregate / approxQuantile action becomes one GPU reduction +
RCCL scalar all-reduce (utils/stats.py).
"""

from __future__ import annotations

import torch

from .. import persistence
from ..estimator import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    RegressionModel,
    Regressor,
)
from ..frame import TensorFrame
from ..params import Params, ParamValidators
from ..parallel import get_comm
from ..utils.stats import dist_quantile, dist_weighted_mean


class _DummyRegressorParams(Params):
    def _declare_params(self):
        super()._declare_params()
        self.strategy = self._str_param(
            "strategy",
            "one of mean|median|quantile|constant",
            ParamValidators.inArray(["mean", "median", "quantile", "constant"]),
        )
        self.quantile = self._float_param(
            "quantile", "quantile for strategy=quantile", ParamValidators.inRange(0, 1)
        )
        self.constant = self._float_param("constant", "constant for strategy=constant")
        self.tol = self._float_param(
            "tol", "quantile precision (advisory; GPU quantile is near-exact)",
            ParamValidators.gtEq(0.0),
        )
        self._setDefault(strategy="mean", tol=1e-3)

    def getStrategy(self):
        return self.getOrDefault("strategy")

    def setStrategy(self, v):
        return self.set("strategy", v)

    def setQuantile(self, v):
        return self.set("quantile", v)

    def setConstant(self, v):
        return self.set("constant", v)

    def setTol(self, v):
        return self.set("tol", v)


class DummyRegressor(Regressor, _DummyRegressorParams):
    def _fit(self, dataset: TensorFrame) -> "DummyRegressionModel":
        x, y, w = self._extract_xyw(dataset)
        comm = get_comm()
        strategy = self.getStrategy()
        if strategy == "mean":
            c = dist_weighted_mean(y, w, comm)
        elif strategy == "median":
            c = dist_quantile(y, 0.5, w, comm)
        elif strategy == "quantile":
            c = dist_quantile(y, self.getOrDefault("quantile"), w, comm)
        else:
            c = self.getOrDefault("constant")
        model = DummyRegressionModel()
        model._constant = float(c)
        model._num_features = x.shape[1]
        for p in ("featuresCol", "labelCol", "predictionCol"):
            model.set(p, self.getOrDefault(p))
        return model


class DummyRegressionModel(RegressionModel, _DummyRegressorParams):
    _constant: float = 0.0

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        return torch.full(
            (features.shape[0],), self._constant, dtype=torch.float32,
            device=features.device,
        )

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={"constant": self._constant, "numFeatures": self._num_features},
        )

    def _load_extra(self, path: str, meta: dict):
        self._constant = float(meta["constant"])
        self._num_features = meta.get("numFeatures", -1)


class _DummyClassifierParams(Params):
    def _declare_params(self):
        super()._declare_params()
        self.strategy = self._str_param(
            "strategy",
            "one of uniform|prior|constant",
            ParamValidators.inArray(["uniform", "prior", "constant"]),
        )
        self.constant = self._int_param("constant", "class for strategy=constant")
        self._setDefault(strategy="prior")

    def getStrategy(self):
        return self.getOrDefault("strategy")

    def setStrategy(self, v):
        return self.set("strategy", v)

    def setConstant(self, v):
        return self.set("constant", v)


class DummyClassifier(ProbabilisticClassifier, _DummyClassifierParams):
    def _fit(self, dataset: TensorFrame) -> "DummyClassificationModel":
        x, y, w = self._extract_xyw(dataset)
        comm = get_comm()
        k = int(comm.all_reduce_scalar(self._get_num_classes(dataset), "max"))
        strategy = self.getStrategy()
        if strategy == "uniform":
            prob = torch.full((k,), 1.0 / k)
            raw = prob.log()
        elif strategy == "prior":
            # per-class masked sums, NOT index_add_: a 10M-row f64
            # index_add onto k cells is a CAS-loop pileup on gfx950
            # (~2.2 s measured for k=2); k reduction kernels are ~ms
            yl = y.long()
            counts = torch.stack(
                [(w * (yl == c).to(w.dtype)).sum().double() for c in range(k)]
            ).to(x.device)
            comm.all_reduce_(counts)
            prob = (counts / counts.sum()).float().cpu()
            raw = prob.clamp_min(1e-300).log()
        else:
            c = self.getOrDefault("constant")
            prob = torch.zeros(k)
            prob[c] = 1.0
            raw = prob.clamp_min(1e-300).log()
        model = DummyClassificationModel()
        model._raw = raw
        model._prob = prob
        model._num_classes = k
        model._num_features = x.shape[1]
        for p in (
            "featuresCol", "labelCol", "predictionCol",
            "rawPredictionCol", "probabilityCol",
        ):
            model.set(p, self.getOrDefault(p))
        return model


class DummyClassificationModel(
    ProbabilisticClassificationModel, _DummyClassifierParams
):
    """Fixed raw/probability vectors (reference DummyClassifier.scala:163-165).

    ``from_raw`` builds the log-odds binary init variant that GBMClassifier
    constructs directly (reference GBMClassifier.scala:275-283)."""

    _raw: torch.Tensor
    _prob: torch.Tensor

    @classmethod
    def from_raw(cls, raw: torch.Tensor, num_classes: int, num_features: int = -1):
        m = cls()
        m._raw = raw.float().cpu()
        p = raw.float().softmax(0) if raw.numel() > 1 else torch.sigmoid(raw)
        m._prob = p.cpu()
        m._num_classes = num_classes
        m._num_features = num_features
        return m

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        return self._raw.to(features.device).unsqueeze(0).expand(
            features.shape[0], -1
        ).clone()

    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        p = self._prob.to(raw.device).unsqueeze(0).expand(raw.shape[0], -1)
        raw.copy_(p)
        return raw

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self,
            path,
            extra={
                "raw": self._raw.tolist(),
                "prob": self._prob.tolist(),
                "numClasses": self._num_classes,
                "numFeatures": self._num_features,
            },
        )

    def _load_extra(self, path: str, meta: dict):
        self._raw = torch.tensor(meta["raw"], dtype=torch.float32)
        self._prob = torch.tensor(meta["prob"], dtype=torch.float32)
        self._num_classes = meta["numClasses"]
        self._num_features = meta.get("numFeatures", -1)
