"""BoostingRegressor — Drucker AdaBoost.R2
(reference regression/BoostingRegressor.scala:173-285).

Per round: fit on weights normalized by their sum; per-row absolute error;
max-error reduce; normalized losses through the loss map {exponential:
1-e^-e, linear: e, squared: e^2} (:97-106); weighted estimator error; stop
when maxError == 0 or estimatorError >= 0.5 (the fitted model is kept — the
reference's ``best`` bookkeeping nets out to "keep every fitted model");
beta = err/(1-err), stage weight log(1/beta) (1.0 when beta == 0); reweight
w * beta^(1-loss).  Model prediction: weighted MEDIAN (via
Utils.weightedMedian) or weighted mean of per-model predictions (:333-340).

MI355X: per-row error/loss maps are tensor ops on the resident shard; the
three reductions per round (sum, max, weighted error) are RCCL scalar
all-reduces.
"""

from __future__ import annotations

import math
import os
from typing import List

import torch

from .. import persistence
from ..ensemble.binning import BinnedDataset
from ..ensemble.params import (
    HasAggregationDepth,
    HasBaseLearner,
    HasCheckpointInterval,
    HasNumBaseLearners,
)
from ..ensemble.utils import weighted_median
from ..estimator import RegressionModel, Regressor
from ..frame import TensorFrame
from ..params import ParamValidators
from ..parallel import get_comm

LOSS_MAP = {
    "exponential": lambda e: 1.0 - torch.exp(-e),
    "linear": lambda e: e,
    "squared": lambda e: e * e,
}


class _BoostingRegressorParams(
    HasNumBaseLearners, HasBaseLearner, HasCheckpointInterval, HasAggregationDepth
):
    def _declare_params(self):
        super()._declare_params()
        self.lossType = self._str_param(
            "lossType",
            "loss for weight updates: exponential|squared|linear",
            ParamValidators.inArray(list(LOSS_MAP)),
        )
        self.votingStrategy = self._str_param(
            "votingStrategy",
            "median or mean aggregation",
            ParamValidators.inArray(["median", "mean"]),
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(
            numBaseLearners=10, lossType="linear", votingStrategy="median", seed=0
        )

    def getLossType(self):
        return self.getOrDefault("lossType")

    def setLossType(self, v):
        return self.set("lossType", v)

    def getVotingStrategy(self):
        return self.getOrDefault("votingStrategy")

    def setVotingStrategy(self, v):
        return self.set("votingStrategy", v)

    def setSeed(self, v):
        return self.set("seed", v)


class BoostingRegressor(Regressor, _BoostingRegressorParams):
    def _default_base_learner(self):
        from ..models.tree import DecisionTreeRegressor

        return DecisionTreeRegressor()

    def _fit(self, dataset: TensorFrame) -> "BoostingRegressionModel":
        from ..utils.instrumentation import Instrumentation

        instr = Instrumentation(self, dataset)
        instr.log_params(self, "lossType", "numBaseLearners", "votingStrategy")
        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        k = self.getNumBaseLearners()
        loss_fn = LOSS_MAP[self.getLossType()]
        x, y, w = self._extract_xyw(dataset)
        binned = BinnedDataset(x, dataset)

        boosting_w = w.clone()
        sum_w = comm.all_reduce_scalar(float(boosting_w.sum()))

        # resume: replay saved models through the reweighting loop (same
        # stats recomputed, no refit) — SURVEY.md §5.4
        from ..utils import checkpoint as ckpt

        ckpt_dir = self.getCheckpointDir()
        ck_fp = (
            ckpt.fingerprint(self, x.shape[0], x.shape[1], y, w)
            if ckpt_dir else None
        )
        ck_saved: set = set()
        resumed = ckpt.load_round_state(ckpt_dir, ck_fp)
        replay = resumed[1][:k] if resumed else []
        if replay:
            instr.log_named_value("resumed_from_round", len(replay))

        models: List = []
        est_weights: List[float] = []
        i = 0
        done = False
        while i < k and not done and sum_w > 0:
            norm_w = boosting_w / sum_w
            if i < len(replay):
                model = replay[i]
            else:
                fr = binned.fit_frame(learner, y, norm_w)
                model = self.fit_base_learner(learner, fr, weight_col="weight")

            errors = (y - model.predict(x)).abs()
            max_err = comm.all_reduce_scalar(
                float(errors.max()) if errors.numel() else 0.0, "max"
            )
            if max_err == 0:
                done = True
                losses = loss_fn(errors)
            else:
                losses = loss_fn(errors / max_err)

            est_err = comm.all_reduce_scalar(float((norm_w * losses).sum()))
            if est_err >= 0.5:
                done = True
            beta = est_err / (1.0 - est_err) if est_err < 1.0 else float("inf")
            est_weight = 1.0 if beta == 0.0 else math.log(1.0 / beta)

            boosting_w = norm_w * torch.pow(
                torch.tensor(beta, device=x.device), 1.0 - losses
            )
            sum_w = comm.all_reduce_scalar(float(boosting_w.sum()))

            models.append(model)
            est_weights.append(est_weight)
            instr.log_round(i, error=est_err, weight=est_weight, sum_w=sum_w)
            interval = self.getCheckpointInterval()
            if (ckpt_dir and i >= len(replay) and interval > 0
                    and (i + 1) % interval == 0):
                ckpt.save_round_state(ckpt_dir, i + 1, models, est_weights,
                                      fingerprint=ck_fp, _saved_dirs=ck_saved)
            i += 1

        instr.finish()
        ckpt.clear(ckpt_dir)  # resume state is crash recovery only
        model = BoostingRegressionModel()
        model._models = models
        model._weights = est_weights
        model._num_features = x.shape[1]
        model.set("votingStrategy", self.getVotingStrategy())
        for p in ("featuresCol", "labelCol", "predictionCol"):
            model.set(p, self.getOrDefault(p))
        return model

    def _save_impl(self, path: str):
        persistence.save_metadata(self, path)
        self._save_learner(path)

    def _load_extra(self, path: str, meta: dict):
        self.setBaseLearner(self._load_learner(path))


class BoostingRegressionModel(RegressionModel, _BoostingRegressorParams):
    @property
    def models(self):
        """Reference BoostingRegressionModel.models."""
        return list(self._models)

    @property
    def weights(self):
        """Per-stage estimator weights (reference .weights)."""
        return list(self._weights)
    _models: List = []
    _weights: List[float] = []

    @property
    def numModels(self):
        return len(self._models)

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        x = features.float()
        preds = torch.stack([m.predict(x) for m in self._models], dim=1)  # [N, M]
        wts = torch.tensor(self._weights, dtype=torch.float32, device=x.device)
        if self.getVotingStrategy() == "median":
            return weighted_median(preds, wts.unsqueeze(0).expand_as(preds))
        return (preds * wts).sum(dim=1) / wts.sum()

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={"numModels": len(self._models), "numFeatures": self._num_features},
        )
        for i, m in enumerate(self._models):
            m.save(os.path.join(path, f"model-{i}"), overwrite=True)
            persistence.save_json_rows(
                os.path.join(path, f"data-{i}"), [{"weight": self._weights[i]}]
            )

    def _load_extra(self, path: str, meta: dict):
        self._num_features = meta.get("numFeatures", -1)
        self._models = []
        self._weights = []
        i = 0
        while os.path.isdir(os.path.join(path, f"model-{i}")):
            self._models.append(
                persistence.load_instance(os.path.join(path, f"model-{i}"))
            )
            self._weights.append(
                persistence.load_json_rows(os.path.join(path, f"data-{i}"))[0]["weight"]
            )
            i += 1
