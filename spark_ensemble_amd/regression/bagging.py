"""BaggingRegressor — bootstrap aggregation with feature subspaces (SubBag).

Re-creates reference regression/BaggingRegressor.scala:117-290: per learner
i, a row resample (Poisson/Bernoulli as weights — the tensor-native form of
``RDD.sample`` at :149-150), a feature subspace ``subspace(ratio, nF,
seed+i)`` (:141-143), and a base-learner fit; the model predicts the MEAN of
per-model predictions on sliced features (:221-228).  Persistence uses the
``model-<i>/`` + ``data-<i>/`` (JSON {subspace}) layout (:245-290).

Documented deviation: the reference row-samples with the same seed for
every learner (only the subspace varies, a known quirk of :149-150); we use
seed + i so bags are independent (what bagging intends).

MI355X notes: fits share one BinnedDataset (features binned once); weight-
based sampling keeps the HBM-resident features un-copied; independent fits
are dispatched per ``parallelism`` (HIP streams share the GPU naturally —
each tree level already saturates the chip, so sequential fits lose nothing
at large N).
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
from ..estimator import RegressionModel, Regressor
from ..frame import TensorFrame
from ..parallel import get_comm


class _BaggingRegressorParams(
    HasNumBaseLearners, HasBaseLearner, HasParallelism, HasSubBag
):
    def _declare_params(self):
        super()._declare_params()
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(numBaseLearners=10, seed=0)

    def setSeed(self, v):
        return self.set("seed", v)


class BaggingRegressor(Regressor, _BaggingRegressorParams):
    def _default_base_learner(self):
        from ..models.tree import DecisionTreeRegressor

        return DecisionTreeRegressor()

    def _fit(self, dataset: TensorFrame) -> "BaggingRegressionModel":
        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        seed = self.getOrDefault("seed")
        k = self.getNumBaseLearners()
        x, y, w = self._extract_xyw(dataset)
        n, num_features = x.shape
        binned = BinnedDataset(x, dataset)

        subspaces = [
            subspace(self.getSubspaceRatio(), num_features, seed + i) for i in range(k)
        ]

        from ..models.tree import DecisionTreeRegressor, fit_tree_forest

        if (
            type(learner) is DecisionTreeRegressor
            and learner.getOrDefault("minWeightFractionPerNode") == 0.0
        ):
            # fused path: all k bagged trees grow level-synchronously in
            # shared launches (one histogram build + one all-reduce per
            # level for the whole ensemble); subspaces become split-search
            # feature masks, bags become per-tree root row sets — exact
            # parity with k sliced weight-masked fits
            # (tree_grower.grow_forest; reference futures analog
            # BaggingRegressor.scala:145-166)
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
            models, _ = fit_tree_forest(
                learner, f_edges, f_bins, y.unsqueeze(1), bag_w, comm,
                subspaces=subspaces, root_rows=root_rows,
            )
        else:
            if learner is not None and learner.hasParam("maxBins"):
                binned.get(int(learner.getOrDefault("maxBins")))  # pre-warm

            def fit_one(i):
                def task():
                    bw = self.sample_weights(
                        self.getReplacement(),
                        self.getSubsampleRatio(),
                        n,
                        seed + i,
                        x.device,
                        w,
                        comm.rank,
                    )
                    fr = binned.fit_frame(learner, y, bw, subspaces[i])
                    return self.fit_base_learner(learner, fr, weight_col="weight")
                return task

            from ..parallel.streams import parallel_fits

            models = parallel_fits([fit_one(i) for i in range(k)],
                                   self.getParallelism())

        model = BaggingRegressionModel()
        model._models = models
        model._subspaces = subspaces
        model._num_features = num_features
        for p in ("featuresCol", "labelCol", "predictionCol"):
            model.set(p, self.getOrDefault(p))
        return model

    # ---- fold-vectorized fitting (OOF stacking fast path) ----------------
    def _can_fit_folds(self) -> bool:
        from ..models.tree import DecisionTreeRegressor

        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        return (
            type(learner) is DecisionTreeRegressor
            and learner.getOrDefault("minWeightFractionPerNode") == 0.0
        )

    def _fit_folds(self, dataset: TensorFrame, fold: torch.Tensor,
                   num_folds: int, include_full: bool = False):
        """num_folds leave-one-fold-out bagging fits as ONE fused forest:
        every (fold, member) tree gets its own bag row set (bag weight > 0
        AND in-fold) and its member's subspace mask — num_folds * k trees
        grow level-synchronously.  Exact parity with per-fold sequential
        fits: members reuse the same seed + i bags/subspaces per fold,
        just as repeated `fit` calls would."""
        from ..models.tree import fit_tree_forest

        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        seed = self.getOrDefault("seed")
        k = self.getNumBaseLearners()
        x, y, w = self._extract_xyw(dataset)
        n, num_features = x.shape
        binned = BinnedDataset(x, dataset)
        edges, bins = binned.get(int(learner.getOrDefault("maxBins")))
        fold = fold.to(x.device)

        subspaces = [
            subspace(self.getSubspaceRatio(), num_features, seed + i)
            for i in range(k)
        ]
        bags = [
            self.sample_weights(
                self.getReplacement(), self.getSubsampleRatio(), n,
                seed + i, x.device, w, comm.rank,
            )
            for i in range(k)
        ]
        cols = []
        subs_T = []
        for f in range(num_folds):
            mask = (fold != f).float()
            for i in range(k):
                cols.append(bags[i] * mask)
                subs_T.append(subspaces[i])
        n_groups = num_folds
        if include_full:
            # the final full-data refit rides in the same fused forest
            for i in range(k):
                cols.append(bags[i])
                subs_T.append(subspaces[i])
            n_groups += 1
        w_T = torch.stack(cols, dim=1)
        root_rows = [
            (w_T[:, t] > 0).nonzero(as_tuple=True)[0].to(torch.int32)
            for t in range(w_T.shape[1])
        ]
        models, _ = fit_tree_forest(
            learner, edges, bins, y.unsqueeze(1), w_T, comm,
            subspaces=subs_T, root_rows=root_rows,
        )
        out = []
        for f in range(n_groups):
            model = BaggingRegressionModel()
            model._models = models[f * k:(f + 1) * k]
            model._subspaces = subspaces
            model._num_features = num_features
            for p in ("featuresCol", "labelCol", "predictionCol"):
                model.set(p, self.getOrDefault(p))
            out.append(model)
        if include_full:
            return out[:-1], out[-1]
        return out

    def _save_impl(self, path: str):
        persistence.save_metadata(self, path)
        self._save_learner(path)

    def _load_extra(self, path: str, meta: dict):
        self.setBaseLearner(self._load_learner(path))


class BaggingRegressionModel(RegressionModel, _BaggingRegressorParams):
    @property
    def models(self):
        """Fitted member models (reference BaggingRegressionModel.models)."""
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

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        from ..ensemble.utils import packed_forest_margin

        x = features.float()
        mcount = len(self._models)
        packed = packed_forest_margin(
            x, self._models, [1.0 / mcount] * mcount, self._subspaces,
            x.shape[1],
            cache=self.__dict__.setdefault("_pack_cache", {}),
        )
        if packed is not None:
            return packed
        acc = None
        for sub, m in zip(self._subspaces, self._models):
            p = m.predict(slice_features(x, sub))
            acc = p if acc is None else acc + p
        return acc / mcount

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={"numModels": len(self._models), "numFeatures": self._num_features},
        )
        for i, m in enumerate(self._models):
            m.save(os.path.join(path, f"model-{i}"), overwrite=True)
            persistence.save_json_rows(
                os.path.join(path, f"data-{i}"),
                [{"subspace": self._subspaces[i].tolist()}],
            )

    def _load_extra(self, path: str, meta: dict):
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
