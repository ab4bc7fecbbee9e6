"""StackingClassifier (reference classification/StackingClassifier.scala).

stackMethod (reference :60-74, default "class"): per base model the
meta-features are the scalar prediction ("class"), the raw margin vector
("raw") or the probability vector ("proba") — so the meta-feature width is
M or M*K (:190-202).  Like the reference, the fitted model extends the
plain prediction surface (``transform`` emits prediction only).

Out-of-fold mode as in StackingRegressor (see its docstring);
``inSample=True`` restores reference behavior.
"""

from __future__ import annotations

import os
from typing import List

import torch

from .. import persistence
from ..ensemble.params import HasBaseLearners, HasParallelism, HasStacker
from ..estimator import PredictionModel, Predictor
from ..frame import TensorFrame
from ..params import ParamValidators
from ..parallel import get_comm
from .bagging import ProbabilisticClassificationModel  # noqa: F401  (type ref)
from ..regression.stacking import _check_weight_support


class _StackingClassifierParams(HasBaseLearners, HasStacker, HasParallelism):
    def _declare_params(self):
        super()._declare_params()
        self.stackMethod = self._str_param(
            "stackMethod",
            "meta-feature source: class|raw|proba",
            ParamValidators.inArray(["class", "raw", "proba"]),
        )
        self.numFolds = self._int_param(
            "numFolds", "folds for out-of-fold stacking", ParamValidators.gtEq(2)
        )
        self.inSample = self._bool_param(
            "inSample", "use reference-style in-sample stacking (no OOF)"
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(stackMethod="class", numFolds=5, inSample=False, seed=0)

    def getStackMethod(self):
        return self.getOrDefault("stackMethod")

    def setStackMethod(self, v):
        return self.set("stackMethod", v)

    def setNumFolds(self, v):
        return self.set("numFolds", v)

    def setInSample(self, v):
        return self.set("inSample", v)

    def setSeed(self, v):
        return self.set("seed", v)


def _model_meta(m, x, method):
    if method == "proba" and hasattr(m, "predictProbability"):
        return m.predictProbability(x)
    if method == "raw" and hasattr(m, "predictRaw"):
        return m.predictRaw(x)
    return m.predict(x).unsqueeze(1)


def _meta_features_clf(models, x, method):
    return torch.cat([_model_meta(m, x, method) for m in models], dim=1)


class StackingClassifier(Predictor, _StackingClassifierParams):
    def _fit(self, dataset: TensorFrame) -> "StackingClassificationModel":
        comm = get_comm()
        learners = self.getBaseLearners()
        stacker = self.getStacker()
        method = self.getStackMethod()
        x, y, w = self._extract_xyw(dataset)
        n = x.shape[0]
        wcol = _check_weight_support(learners, stacker, self.getWeightCol())
        use_w = w if wcol else torch.ones_like(w)

        from ..parallel.streams import parallel_fits

        # shared derived-data cache: features binned once for every
        # fold/final fit (see regression/stacking.py)
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
            meta = _meta_features_clf(models, x, method)
        else:
            num_folds = self.getOrDefault("numFolds")
            g = torch.Generator().manual_seed(self.getOrDefault("seed"))
            fold = torch.randint(0, num_folds, (n,), generator=g).to(x.device)

            # fold-vectorized learners (default-config GBM classifiers)
            # grow all their fold models jointly — see regression/stacking
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
            meta_cols: List[torch.Tensor] = []
            for mi, lr in enumerate(learners):
                col = None
                for f in range(num_folds):
                    m = fold_models[mi][f]
                    sel = fold == f
                    part = _model_meta(m, x[sel], method)
                    if col is None:
                        col = torch.zeros(
                            n, part.shape[1], dtype=torch.float32, device=x.device
                        )
                    col[sel] = part
                meta_cols.append(col)
            meta = torch.cat(meta_cols, dim=1)
            models = [final_models[mi] for mi in range(len(learners))]

        stack = self.fit_base_learner(
            stacker,
            TensorFrame(features=meta, label=y, weight=use_w),
            weight_col="weight",
        )

        model = StackingClassificationModel()
        model._models = models
        model._stack = stack
        model._num_features = x.shape[1]
        model.set("stackMethod", method)
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


class StackingClassificationModel(PredictionModel, _StackingClassifierParams):
    @property
    def models(self):
        """Fitted base models (reference field)."""
        return list(self._models)

    @property
    def stack(self):
        return self._stack
    _models: List = []
    _stack = None

    @property
    def numModels(self):
        return len(self._models)

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        x = features.float()
        meta = _meta_features_clf(self._models, x, self.getStackMethod())
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
