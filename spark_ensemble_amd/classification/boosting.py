"""BoostingClassifier — SAMME / SAMME.R AdaBoost
(reference classification/BoostingClassifier.scala:135-282).

algorithm = "real" (SAMME.R): estimator weight 1.0; reweight
w * exp(-((K-1)/K) * sum_c code_c * log max(p_c, EPS)) with code = 1 for the
true class else -1/(K-1) (:218-227); done when weighted misclassification
<= 0.

algorithm = "discrete" (SAMME): weighted 0/1 error; beta =
err/((1-err)(K-1)); weight log(1/beta) (1.0 if beta == 0); if
err >= 1 - 1/K the just-fitted learner is REVERTED (reference :252
``i = i - 1; done = true``); reweight w * (1/beta)^err.

Model (:334-382): SAMME.R decision sums (K-1)(log p - mean log p) per
model; SAMME sums +-weight votes (+w on the predicted class, -w/(K-1)
elsewhere); raw2probability = softmax(raw / (K-1)) (:342-346).
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
from ..estimator import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
)
from ..frame import TensorFrame
from ..params import ParamValidators
from ..parallel import get_comm

EPSILON = 2.220446049250313e-16  # Spark ml.impl.Utils.EPSILON


class _BoostingClassifierParams(
    HasNumBaseLearners, HasBaseLearner, HasCheckpointInterval, HasAggregationDepth
):
    def _declare_params(self):
        super()._declare_params()
        self.algorithm = self._str_param(
            "algorithm",
            "real (SAMME.R) or discrete (SAMME)",
            ParamValidators.inArray(["real", "discrete"]),
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(numBaseLearners=10, algorithm="real", seed=0)

    def getAlgorithm(self):
        return self.getOrDefault("algorithm")

    def setAlgorithm(self, v):
        return self.set("algorithm", v)

    def setSeed(self, v):
        return self.set("seed", v)


class BoostingClassifier(ProbabilisticClassifier, _BoostingClassifierParams):
    def _default_base_learner(self):
        from ..models.tree import DecisionTreeClassifier

        return DecisionTreeClassifier()

    def _fit(self, dataset: TensorFrame) -> "BoostingClassificationModel":
        from ..utils.instrumentation import Instrumentation

        instr = Instrumentation(self, dataset)
        instr.log_params(self, "algorithm", "numBaseLearners")
        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        k = self.getNumBaseLearners()
        algo = self.getAlgorithm()
        x, y, w = self._extract_xyw(dataset)
        num_classes = int(comm.all_reduce_scalar(self._get_num_classes(dataset), "max"))
        kc = float(num_classes)
        binned = BinnedDataset(x, dataset)
        yl = y.long()

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

            if algo == "real":
                if not hasattr(model, "predictProbability"):
                    raise RuntimeError(
                        'algorithm "real" needs a probabilistic base learner '
                        "(reference BoostingClassifier.scala:261-263)"
                    )
                prob = model.predictProbability(x)  # [N, K]
                mispred = (prob.argmax(dim=1) != yl).float()
                est_err = comm.all_reduce_scalar(float((norm_w * mispred).sum()))
                if est_err <= 0:
                    done = True
                models.append(model)
                est_weights.append(1.0)
                # SAMME.R reweight (:218-227)
                logp = prob.clamp_min(EPSILON).log()
                code = torch.full_like(logp, -1.0 / (kc - 1.0))
                code.scatter_(1, yl.unsqueeze(1), 1.0)
                loss = (code * logp).sum(dim=1)
                boosting_w = norm_w * torch.exp(-((kc - 1.0) / kc) * loss)
            else:
                pred = model.predict(x)
                err01 = (pred != y).float()
                est_err = comm.all_reduce_scalar(float((norm_w * err01).sum()))
                if est_err <= 0:
                    done = True
                beta = est_err / ((1.0 - est_err) * (kc - 1.0)) if est_err < 1.0 else float("inf")
                est_weight = 1.0 if beta == 0.0 else math.log(1.0 / beta)
                models.append(model)
                est_weights.append(est_weight)
                if est_err >= 1.0 - 1.0 / kc:
                    # revert this learner (reference :252)
                    models.pop()
                    est_weights.pop()
                    done = True
                inv_beta = float("inf") if beta == 0.0 else 1.0 / beta
                boosting_w = norm_w * torch.pow(
                    torch.tensor(inv_beta, device=x.device), err01
                )
            sum_w = comm.all_reduce_scalar(float(boosting_w.sum()))
            instr.log_round(i, error=est_err,
                            weight=est_weights[-1] if est_weights else 0.0,
                            sum_w=sum_w)
            interval = self.getCheckpointInterval()
            if (ckpt_dir and i >= len(replay) and interval > 0
                    and (i + 1) % interval == 0):
                ckpt.save_round_state(ckpt_dir, i + 1, models, est_weights,
                                      fingerprint=ck_fp, _saved_dirs=ck_saved)
            i += 1

        instr.finish()
        ckpt.clear(ckpt_dir)  # resume state is crash recovery only
        model = BoostingClassificationModel()
        model._models = models
        model._weights = est_weights
        model._num_classes = num_classes
        model._num_features = x.shape[1]
        model.set("algorithm", algo)
        for p in (
            "featuresCol", "labelCol", "predictionCol",
            "rawPredictionCol", "probabilityCol",
        ):
            model.set(p, self.getOrDefault(p))
        return model

    def _save_impl(self, path: str):
        persistence.save_metadata(self, path)
        self._save_learner(path)

    def _load_extra(self, path: str, meta: dict):
        self.setBaseLearner(self._load_learner(path))


class BoostingClassificationModel(
    ProbabilisticClassificationModel, _BoostingClassifierParams
):
    # public accessors (reference BoostingClassificationModel fields)
    @property
    def models(self):
        return list(self._models)

    @property
    def weights(self):
        return list(self._weights)
    _models: List = []
    _weights: List[float] = []

    @property
    def numModels(self):
        return len(self._models)

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        x = features.float()
        k = self._num_classes
        kc = float(k)
        res = torch.zeros(x.shape[0], k, dtype=torch.float32, device=x.device)
        if self.getAlgorithm() == "real":
            for m in self._models:
                logp = m.predictProbability(x).clamp_min(EPSILON).log()
                dec = logp - logp.mean(dim=1, keepdim=True)
                res += (kc - 1.0) * dec
        else:
            for wgt, m in zip(self._weights, self._models):
                pred = m.predict(x).long()
                res -= wgt / (kc - 1.0)
                res.scatter_add_(
                    1, pred.unsqueeze(1),
                    torch.full((x.shape[0], 1), wgt * (1.0 + 1.0 / (kc - 1.0)),
                               device=x.device),
                )
        return res

    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        raw /= (self._num_classes - 1.0)
        return torch.softmax(raw, dim=1, out=raw)

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={
                "numClasses": self._num_classes,
                "numModels": len(self._models),
                "numFeatures": self._num_features,
            },
        )
        for i, m in enumerate(self._models):
            m.save(os.path.join(path, f"model-{i}"), overwrite=True)
            persistence.save_json_rows(
                os.path.join(path, f"data-{i}"), [{"weight": self._weights[i]}]
            )

    def _load_extra(self, path: str, meta: dict):
        self._num_classes = meta["numClasses"]
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
