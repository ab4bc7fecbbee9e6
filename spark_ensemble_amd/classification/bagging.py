"""BaggingClassifier (reference classification/BaggingClassifier.scala).

Voting (reference :55-67, 260-287): ``soft`` sums per-model
``predictProbability``; ``hard`` sums one-hot votes of per-model ``predict``
(default hard, :67); soft voting with a non-probabilistic base model raises
(:275-277).  ``raw2probability`` divides by numModels (:285-287).
Same per-learner subspace seed+i scheme and seed deviation as
BaggingRegressor (see its docstring).
"""

from __future__ import annotations

import os
from typing import List

import torch

from .. import persistence
from ..ensemble.binning import BinnedDataset
from ..ensemble.params import (
    HasBaseLearner,
    HasNumBaseLearners,
    HasParallelism,
    HasSubBag,
)
from ..ensemble.utils import slice_features, subspace
from ..estimator import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
)
from ..frame import TensorFrame
from ..params import ParamValidators
from ..parallel import get_comm


class _BaggingClassifierParams(
    HasNumBaseLearners, HasBaseLearner, HasParallelism, HasSubBag
):
    def _declare_params(self):
        super()._declare_params()
        self.votingStrategy = self._str_param(
            "votingStrategy",
            "hard (majority vote) or soft (mean probability)",
            ParamValidators.inArray(["hard", "soft"]),
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(numBaseLearners=10, votingStrategy="hard", seed=0)

    def getVotingStrategy(self):
        return self.getOrDefault("votingStrategy")

    def setVotingStrategy(self, v):
        return self.set("votingStrategy", v)

    def setSeed(self, v):
        return self.set("seed", v)


class BaggingClassifier(ProbabilisticClassifier, _BaggingClassifierParams):
    def _default_base_learner(self):
        from ..models.tree import DecisionTreeClassifier

        return DecisionTreeClassifier()

    def _fit(self, dataset: TensorFrame) -> "BaggingClassificationModel":
        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        seed = self.getOrDefault("seed")
        k = self.getNumBaseLearners()
        x, y, w = self._extract_xyw(dataset)
        n, num_features = x.shape
        num_classes = int(comm.all_reduce_scalar(self._get_num_classes(dataset), "max"))
        binned = BinnedDataset(x, dataset)

        subspaces = [
            subspace(self.getSubspaceRatio(), num_features, seed + i) for i in range(k)
        ]

        from ..models.tree import DecisionTreeClassifier, fit_class_tree_forest

        if (
            type(learner) is DecisionTreeClassifier
            and learner.getOrDefault("minWeightFractionPerNode") == 0.0
            and num_classes + 2 <= 8
        ):
            # fused path: all k gini trees (K one-hot channels each) grow
            # level-synchronously; subspaces as split masks, bags as root
            # row sets — same machinery as the regression fusion
            f_edges, f_bins = binned.get(int(learner.getOrDefault("maxBins")))
            bag_w = torch.stack([
                self.sample_weights(
                    self.getReplacement(), self.getSubsampleRatio(), n,
                    seed + i, x.device, w, comm.rank,
                )
                for i in range(k)
            ], dim=1)
            root_rows = [
                (bag_w[:, i] > 0).nonzero(as_tuple=True)[0].to(torch.int32)
                for i in range(k)
            ]
            onehot = torch.zeros(n, num_classes, dtype=torch.float32,
                                 device=x.device)
            onehot.scatter_(1, y.long().unsqueeze(1), 1.0)
            models = fit_class_tree_forest(
                learner, f_edges, f_bins, onehot, bag_w, num_classes, comm,
                subspaces=subspaces, root_rows=root_rows,
            )
            model = BaggingClassificationModel()
            model._models = models
            model._subspaces = subspaces
            model._num_classes = num_classes
            model._num_features = num_features
            model.set("votingStrategy", self.getVotingStrategy())
            for p in (
                "featuresCol", "labelCol", "predictionCol",
                "rawPredictionCol", "probabilityCol",
            ):
                model.set(p, self.getOrDefault(p))
            return model

        if learner is not None and learner.hasParam("maxBins"):
            binned.get(int(learner.getOrDefault("maxBins")))  # pre-warm once

        def fit_one(i):
            def task():
                bag_w = self.sample_weights(
                    self.getReplacement(),
                    self.getSubsampleRatio(),
                    n,
                    seed + i,
                    x.device,
                    w,
                    comm.rank,
                )
                fr = binned.fit_frame(learner, y, bag_w, subspaces[i])
                return self.fit_base_learner(learner, fr, weight_col="weight")
            return task

        from ..parallel.streams import parallel_fits

        models = parallel_fits([fit_one(i) for i in range(k)],
                               self.getParallelism())

        model = BaggingClassificationModel()
        model._models = models
        model._subspaces = subspaces
        model._num_classes = num_classes
        model._num_features = num_features
        model.set("votingStrategy", self.getVotingStrategy())
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


class BaggingClassificationModel(
    ProbabilisticClassificationModel, _BaggingClassifierParams
):
    @property
    def models(self):
        return list(self._models)

    @property
    def subspaces(self):
        return list(self._subspaces)
    _models: List = []
    _subspaces: List[torch.Tensor] = []

    @property
    def numModels(self):
        return len(self._models)

    @property
    def featureImportances(self):
        from ..ensemble.utils import ensemble_feature_importances

        return ensemble_feature_importances(
            self._models, [1.0] * len(self._models), self._subspaces, self._num_features
        )

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        from ..ensemble.utils import packed_forest_vote

        x = features.float()
        k = self._num_classes
        soft = self.getVotingStrategy() == "soft"
        # fast path: every member a built-in classification tree -> ONE
        # packed forest kernel with per-node leaf transforms (normalized
        # probs for soft, one-hot argmax for hard) — exact
        packed = packed_forest_vote(
            x, self._models, self._subspaces, x.shape[1], soft,
            cache=self.__dict__.setdefault(
                "_pack_cache_soft" if soft else "_pack_cache_hard", {}),
        )
        if packed is not None:
            return packed
        acc = torch.zeros(x.shape[0], k, dtype=torch.float32, device=x.device)
        for sub, m in zip(self._subspaces, self._models):
            xs = slice_features(x, sub)
            if soft:
                if not hasattr(m, "predictProbability"):
                    raise RuntimeError(
                        "soft voting requires probabilistic base models "
                        "(reference BaggingClassifier.scala:275-277)"
                    )
                acc += m.predictProbability(xs)
            else:
                pred = m.predict(xs).long()
                acc.scatter_add_(
                    1, pred.unsqueeze(1),
                    torch.ones(x.shape[0], 1, device=x.device),
                )
        return acc

    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        raw /= len(self._models)
        return raw

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
                os.path.join(path, f"data-{i}"),
                [{"subspace": self._subspaces[i].tolist()}],
            )

    def _load_extra(self, path: str, meta: dict):
        self._num_classes = meta["numClasses"]
        self._num_features = meta.get("numFeatures", -1)
        self._models = []
        self._subspaces = []
        i = 0
        while os.path.isdir(os.path.join(path, f"model-{i}")):
            self._models.append(
                persistence.load_instance(os.path.join(path, f"model-{i}"))
            )
            row = persistence.load_json_rows(os.path.join(path, f"data-{i}"))[0]
            self._subspaces.append(torch.tensor(row["subspace"], dtype=torch.long))
            i += 1
