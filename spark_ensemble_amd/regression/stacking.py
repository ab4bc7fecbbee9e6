"""StackingRegressor — stacked generalization
(reference regression/StackingRegressor.scala:104-280).

Reference semantics: all base learners fit in parallel on the FULL dataset;
meta-features are the per-model predictions on the SAME training rows
(in-sample stacking — :155-163); the stacker fits on those; instance
weights honored only when every base learner supports them (:112-119);
model predict = stack.predict([f_1(x) ... f_M(x)]) (:224-226); persistence
adds a ``stack/`` subdir (:247-256).

This rebuild makes k-fold OUT-OF-FOLD stacking the primary mode
(numFolds >= 2, default 5): each base learner is fit numFolds times with
fold-held-out rows weight-masked to zero (no data movement — the resident
feature tensor is untouched), its held-out predictions form the
meta-feature matrix, and the final saved base models are refit on all rows.
``inSample=True`` restores exact reference behavior for parity tests.
"""

from __future__ import annotations

import os
import warnings
from typing import List

import torch

from .. import persistence
from ..ensemble.params import HasBaseLearners, HasParallelism, HasStacker
from ..estimator import RegressionModel, Regressor
from ..frame import TensorFrame
from ..params import ParamValidators
from ..parallel import get_comm


class _StackingRegressorParams(HasBaseLearners, HasStacker, HasParallelism):
    def _declare_params(self):
        super()._declare_params()
        self.numFolds = self._int_param(
            "numFolds", "folds for out-of-fold stacking", ParamValidators.gtEq(2)
        )
        self.inSample = self._bool_param(
            "inSample", "use reference-style in-sample stacking (no OOF)"
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(numFolds=5, inSample=False, seed=0)

    def setNumFolds(self, v):
        return self.set("numFolds", v)

    def setInSample(self, v):
        return self.set("inSample", v)

    def setSeed(self, v):
        return self.set("seed", v)


def _check_weight_support(learners, stacker, weight_col):
    """Weight honored only if every learner supports it (reference
    StackingRegressor.scala:112-119)."""
    if not weight_col:
        return None
    all_support = all(
        lr.hasParam("weightCol") for lr in list(learners) + [stacker]
    )
    if not all_support:
        warnings.warn(
            "weightCol ignored: not every base learner supports instance weights"
        )
        return None
    return weight_col


def _meta_features_reg(models, x):
    return torch.stack([m.predict(x) for m in models], dim=1)


class StackingRegressor(Regressor, _StackingRegressorParams):
    def _fit(self, dataset: TensorFrame) -> "StackingRegressionModel":
        comm = get_comm()
        learners = self.getBaseLearners()
        stacker = self.getStacker()
        x, y, w = self._extract_xyw(dataset)
        n = x.shape[0]
        wcol = _check_weight_support(learners, stacker, self.getWeightCol())
        use_w = w if wcol else torch.ones_like(w)

        from ..parallel.streams import parallel_fits

        # ONE frame whose derived-data cache (binned features) is shared
        # by every fold/final fit below — withColumn propagates the cache,
        # so the features tensor is quantile-binned once per maxBins for
        # the WHOLE stacking fit instead of once per member fit
        shared = TensorFrame(features=x, label=y, weight=use_w)

        if self.getOrDefault("inSample"):
            models = parallel_fits(
                [
                    (lambda lr=lr: self.fit_base_learner(
                        lr, shared, weight_col="weight"))
                    for lr in learners
                ],
                self.getParallelism(), warm_first=True,
            )
            meta = _meta_features_reg(models, x)
        else:
            num_folds = self.getOrDefault("numFolds")
            # identical seeding on every rank: the fold ASSIGNMENT function
            # is shared, the row draws differ because each rank holds
            # different rows of the global dataset
            g = torch.Generator().manual_seed(self.getOrDefault("seed"))
            fold = torch.randint(
                0, num_folds, (n,), generator=g
            ).to(x.device)
            meta = torch.zeros(n, len(learners), dtype=torch.float32, device=x.device)

            # learners with a fold-vectorized fit grow all their fold
            # models JOINTLY (one fused forest per boosting round —
            # GBMRegressor._fit_folds); the rest run per-fold fits on the
            # stream pool
            fused_ok = bool((use_w > 0).all())
            fused = {
                mi for mi, lr in enumerate(learners)
                if fused_ok and hasattr(lr, "_can_fit_folds")
                and lr._can_fit_folds()
            }
            fold_models: dict = {}
            final_models: dict = {}
            for mi in fused:
                fold_models[mi], final_models[mi] = learners[mi]._fit_folds(
                    shared, fold, num_folds, include_full=True
                )

            def fold_task(lr, f):
                def task():
                    wmask = use_w * (fold != f).float()
                    return self.fit_base_learner(
                        lr, shared.withColumn("weight", wmask),
                        weight_col="weight",
                    )
                return task

            plan = [(mi, f) for mi, lr in enumerate(learners)
                    if mi not in fused for f in range(num_folds)]
            tasks = [fold_task(learners[mi], f) for mi, f in plan]
            # final refits for NON-fused learners (fused ones grew theirs
            # inside the fold forest), same pool
            non_fused = [mi for mi in range(len(learners)) if mi not in fused]
            tasks += [
                (lambda lr=learners[mi]: self.fit_base_learner(
                    lr, shared, weight_col="weight"))
                for mi in non_fused
            ]
            fitted = parallel_fits(tasks, self.getParallelism(),
                                   warm_first=True)
            for j, (mi, f) in enumerate(plan):
                fold_models.setdefault(mi, [None] * num_folds)[f] = fitted[j]
            for j, mi in enumerate(non_fused):
                final_models[mi] = fitted[len(plan) + j]
            for mi in range(len(learners)):
                for f in range(num_folds):
                    m = fold_models[mi][f]
                    sel = fold == f
                    meta[sel, mi] = m.predict(x[sel])
            models = [final_models[mi] for mi in range(len(learners))]

        stack = self.fit_base_learner(
            stacker,
            TensorFrame(features=meta, label=y, weight=use_w),
            weight_col="weight",
        )

        model = StackingRegressionModel()
        model._models = models
        model._stack = stack
        model._num_features = x.shape[1]
        for p in ("featuresCol", "labelCol", "predictionCol"):
            model.set(p, self.getOrDefault(p))
        return model

    def _save_impl(self, path: str):
        persistence.save_metadata(self, path)
        self._save_learners(path)
        self._save_stacker(path)

    def _load_extra(self, path: str, meta: dict):
        self.setBaseLearners(self._load_learners(path))
        self.setStacker(self._load_stacker(path))


class StackingRegressionModel(RegressionModel, _StackingRegressorParams):
    @property
    def models(self):
        """Fitted base models (reference StackingRegressionModel.models)."""
        return list(self._models)

    @property
    def stack(self):
        """The meta-learner (reference .stack)."""
        return self._stack
    _models: List = []
    _stack = None

    @property
    def numModels(self):
        return len(self._models)

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        x = features.float()
        meta = _meta_features_reg(self._models, x)
        return self._stack.predict(meta)

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={"numModels": len(self._models), "numFeatures": self._num_features},
        )
        for i, m in enumerate(self._models):
            m.save(os.path.join(path, f"model-{i}"), overwrite=True)
        self._stack.save(os.path.join(path, "stack"), overwrite=True)

    def _load_extra(self, path: str, meta: dict):
        self._num_features = meta.get("numFeatures", -1)
        self._models = []
        i = 0
        while os.path.isdir(os.path.join(path, f"model-{i}")):
            self._models.append(
                persistence.load_instance(os.path.join(path, f"model-{i}"))
            )
            i += 1
        self._stack = persistence.load_instance(os.path.join(path, "stack"))
