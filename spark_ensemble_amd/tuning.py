"""Pipeline / evaluation / model-selection utilities.

The reference's usage docs (reference docs/example.md) drive its
estimators through Spark ML's ``Pipeline``, ``ParamGridBuilder`` and
``CrossValidator``; those live in Spark, not in the reference repo, but a
user switching frameworks needs the same workflow.  This module provides
MI355X-native equivalents over :class:`TensorFrame`:

* :class:`Pipeline` / :class:`PipelineModel` — chain transformers and one
  or more estimators.
* :class:`ParamGridBuilder` — cartesian param grids as override dicts
  (``Estimator.fit(df, params)`` applies them).
* :class:`RegressionEvaluator` (rmse | mse | mae | r2) and
  :class:`MulticlassClassificationEvaluator` (accuracy | f1 |
  weightedPrecision | weightedRecall) — metric computed on the local
  shard then all-reduced, so every rank scores identically.
* :class:`CrossValidator` / :class:`TrainValidationSplit` — k-fold (or
  single-split) selection over a param grid; folds are row masks (no data
  movement), fold assignment is seeded and rank-local.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional, Sequence

import torch

from .estimator import Estimator, Model, Transformer
from .frame import TensorFrame
from .parallel import get_comm


# ---------------------------------------------------------------------------
# Pipeline
# ---------------------------------------------------------------------------


class Pipeline(Estimator):
    """``Pipeline(stages=[t1, est]).fit(df)`` fits estimator stages in
    order, transforming the running frame through each fitted stage.
    Persists with the Spark layout: ``stage-<i>/`` nested saves."""

    def __init__(self, uid=None, stages: Optional[Sequence] = None):
        super().__init__(uid)
        self._stages = list(stages or [])

    def setStages(self, stages):
        self._stages = list(stages)
        return self

    def getStages(self):
        return list(self._stages)

    def _fit(self, dataset: TensorFrame) -> "PipelineModel":
        df = dataset
        fitted: List[Transformer] = []
        for stage in self._stages:
            if isinstance(stage, Estimator):
                model = stage.fit(df)
                fitted.append(model)
                df = model.transform(df)
            elif isinstance(stage, Transformer):
                fitted.append(stage)
                df = stage.transform(df)
            else:
                raise TypeError(f"pipeline stage {stage!r} is not a "
                                "Transformer or Estimator")
        pm = PipelineModel()
        pm._stages = fitted
        return pm

    def _save_impl(self, path: str):
        from . import persistence
        import os

        persistence.save_metadata(self, path,
                                  extra={"numStages": len(self._stages)})
        for i, st in enumerate(self._stages):
            st.save(os.path.join(path, f"stage-{i}"), overwrite=True)

    def _load_extra(self, path: str, meta: dict):
        from . import persistence
        import os

        self._stages = [
            persistence.load_instance(os.path.join(path, f"stage-{i}"))
            for i in range(int(meta.get("numStages", 0)))
        ]


class PipelineModel(Model):
    _stages: List[Transformer]

    @property
    def stages(self):
        return list(self._stages)

    def transform(self, dataset: TensorFrame) -> TensorFrame:
        df = dataset
        for s in self._stages:
            df = s.transform(df)
        return df

    def _save_impl(self, path: str):
        from . import persistence
        import os

        persistence.save_metadata(self, path,
                                  extra={"numStages": len(self._stages)})
        for i, st in enumerate(self._stages):
            st.save(os.path.join(path, f"stage-{i}"), overwrite=True)

    def _load_extra(self, path: str, meta: dict):
        from . import persistence
        import os

        self._stages = [
            persistence.load_instance(os.path.join(path, f"stage-{i}"))
            for i in range(int(meta.get("numStages", 0)))
        ]


# ---------------------------------------------------------------------------
# Param grids
# ---------------------------------------------------------------------------


class ParamGridBuilder:
    def __init__(self):
        self._grid: Dict[str, Sequence] = {}

    def addGrid(self, param: str, values: Sequence) -> "ParamGridBuilder":
        self._grid[str(param)] = list(values)
        return self

    def baseOn(self, **fixed) -> "ParamGridBuilder":
        for k, v in fixed.items():
            self._grid[k] = [v]
        return self

    def build(self) -> List[Dict]:
        maps: List[Dict] = [{}]
        for name, values in self._grid.items():
            maps = [dict(m, **{name: v}) for m in maps for v in values]
        return maps


# ---------------------------------------------------------------------------
# Evaluators
# ---------------------------------------------------------------------------


class _Evaluator:
    def __init__(self, predictionCol="prediction", labelCol="label"):
        self.predictionCol = predictionCol
        self.labelCol = labelCol

    def setPredictionCol(self, v):
        self.predictionCol = v
        return self

    def setLabelCol(self, v):
        self.labelCol = v
        return self

    def evaluate(self, dataset: TensorFrame) -> float:
        raise NotImplementedError

    def isLargerBetter(self) -> bool:
        raise NotImplementedError


class RegressionEvaluator(_Evaluator):
    def __init__(self, metricName="rmse", **kw):
        super().__init__(**kw)
        assert metricName in ("rmse", "mse", "mae", "r2"), metricName
        self.metricName = metricName

    def setMetricName(self, v):
        self.metricName = v
        return self

    def isLargerBetter(self) -> bool:
        return self.metricName == "r2"

    def evaluate(self, dataset: TensorFrame) -> float:
        comm = get_comm()
        y = dataset[self.labelCol].float()
        p = dataset[self.predictionCol].float()
        n = comm.all_reduce_scalar(float(y.numel()))
        if self.metricName in ("rmse", "mse"):
            se = comm.all_reduce_scalar(float(((y - p) ** 2).sum()))
            mse = se / max(n, 1.0)
            return math.sqrt(mse) if self.metricName == "rmse" else mse
        if self.metricName == "mae":
            return comm.all_reduce_scalar(float((y - p).abs().sum())) / max(n, 1.0)
        # r2
        sy = comm.all_reduce_scalar(float(y.sum()))
        mean = sy / max(n, 1.0)
        ss_res = comm.all_reduce_scalar(float(((y - p) ** 2).sum()))
        ss_tot = comm.all_reduce_scalar(float(((y - mean) ** 2).sum()))
        return 1.0 - ss_res / max(ss_tot, 1e-300)


class MulticlassClassificationEvaluator(_Evaluator):
    def __init__(self, metricName="accuracy", **kw):
        super().__init__(**kw)
        assert metricName in (
            "accuracy", "f1", "weightedPrecision", "weightedRecall",
        ), metricName
        self.metricName = metricName

    def setMetricName(self, v):
        self.metricName = v
        return self

    def isLargerBetter(self) -> bool:
        return True

    def evaluate(self, dataset: TensorFrame) -> float:
        comm = get_comm()
        y = dataset[self.labelCol].long()
        p = dataset[self.predictionCol].long()
        if self.metricName == "accuracy":
            n = comm.all_reduce_scalar(float(y.numel()))
            ok = comm.all_reduce_scalar(float((y == p).sum()))
            return ok / max(n, 1.0)
        k = int(comm.all_reduce_scalar(
            float(max(int(y.max()) if y.numel() else 0,
                      int(p.max()) if p.numel() else 0) + 1), "max"))
        # confusion counts: tp, predicted-per-class, actual-per-class
        conf = torch.zeros(3, k, dtype=torch.float64)
        for c in range(k):
            conf[0, c] = float(((y == c) & (p == c)).sum())
            conf[1, c] = float((p == c).sum())
            conf[2, c] = float((y == c).sum())
        if comm.is_distributed:
            dev = conf
            cd = dev.to(y.device) if y.is_cuda else dev
            comm.all_reduce_(cd)
            conf = cd.cpu()
        tp, pred_c, act_c = conf[0], conf[1], conf[2]
        total = float(act_c.sum())
        prec = torch.where(pred_c > 0, tp / pred_c.clamp_min(1), torch.zeros_like(tp))
        rec = torch.where(act_c > 0, tp / act_c.clamp_min(1), torch.zeros_like(tp))
        w = act_c / max(total, 1.0)
        if self.metricName == "weightedPrecision":
            return float((w * prec).sum())
        if self.metricName == "weightedRecall":
            return float((w * rec).sum())
        f1 = torch.where(
            (prec + rec) > 0, 2 * prec * rec / (prec + rec).clamp_min(1e-300),
            torch.zeros_like(prec),
        )
        return float((w * f1).sum())


# ---------------------------------------------------------------------------
# Model selection
# ---------------------------------------------------------------------------


class CrossValidator(Estimator):
    """k-fold cross validation over a param grid.

    Folds are seeded row masks on the local shard; metrics are
    all-reduced inside the evaluator, so every rank selects the same
    winner and refits it on the full data."""

    def __init__(self, uid=None, estimator=None, estimatorParamMaps=None,
                 evaluator=None, numFolds=3, seed=0, parallelism=1):
        super().__init__(uid)
        self.estimator = estimator
        self.estimatorParamMaps = estimatorParamMaps or [{}]
        self.evaluator = evaluator
        self.numFolds = numFolds
        self.seed = seed
        self.parallelism = parallelism

    def setParallelism(self, v):
        self.parallelism = int(v)
        return self

    def setEstimator(self, v):
        self.estimator = v
        return self

    def setEstimatorParamMaps(self, v):
        self.estimatorParamMaps = v
        return self

    def setEvaluator(self, v):
        self.evaluator = v
        return self

    def setNumFolds(self, v):
        self.numFolds = v
        return self

    def setSeed(self, v):
        self.seed = v
        return self

    def _fold_ids(self, n: int, device) -> torch.Tensor:
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        return torch.randint(0, self.numFolds, (n,), generator=g).to(device)

    def _fit(self, dataset: TensorFrame) -> "CrossValidatorModel":
        assert self.estimator is not None and self.evaluator is not None
        folds = self._fold_ids(dataset.count(), dataset.device)

        def scores_fused(pm):
            """Fold-vectorized estimators (default-config GBM / Bagging)
            grow every fold model jointly — CV's leave-one-fold-out IS
            the _fit_folds shape."""
            est = self.estimator.copy(pm) if pm else self.estimator
            if not (hasattr(est, "_can_fit_folds") and est._can_fit_folds()):
                return None
            models = est._fit_folds(dataset, folds, self.numFolds)
            out = []
            for f in range(self.numFolds):
                va = dataset.filter(folds == f)
                out.append(self.evaluator.evaluate(models[f].transform(va)))
            return out

        def cell(pm, f):
            def task():
                tr = dataset.filter(folds != f)
                va = dataset.filter(folds == f)
                model = self.estimator.fit(tr, pm or None)
                return self.evaluator.evaluate(model.transform(va))
            return task

        # (param-map, fold) fits run concurrently on per-thread HIP
        # streams (the Spark CrossValidator parallelism analog); fused
        # param maps skip the pool entirely
        from .parallel.streams import parallel_fits

        fused_scores = {}
        plan = []
        for i, pm in enumerate(self.estimatorParamMaps):
            sc = scores_fused(pm)
            if sc is not None:
                fused_scores[i] = sc
            else:
                plan.append(i)
        tasks = [cell(self.estimatorParamMaps[i], f)
                 for i in plan for f in range(self.numFolds)]
        flat = parallel_fits(tasks, self.parallelism, warm_first=True)
        avg = []
        for i in range(len(self.estimatorParamMaps)):
            if i in fused_scores:
                chunk = fused_scores[i]
            else:
                j = plan.index(i)
                chunk = flat[j * self.numFolds:(j + 1) * self.numFolds]
            avg.append(sum(chunk) / len(chunk))
        better = max if self.evaluator.isLargerBetter() else min
        best_idx = avg.index(better(avg))
        best_model = self.estimator.fit(
            dataset, self.estimatorParamMaps[best_idx] or None
        )
        out = CrossValidatorModel()
        out.bestModel = best_model
        out.avgMetrics = avg
        out.bestIndex = best_idx
        return out


class CrossValidatorModel(Model):
    bestModel: Model
    avgMetrics: List[float]
    bestIndex: int

    def transform(self, dataset: TensorFrame) -> TensorFrame:
        return self.bestModel.transform(dataset)

    def _save_impl(self, path: str):
        import os

        from . import persistence

        persistence.save_metadata(
            self, path,
            extra={"avgMetrics": list(map(float, self.avgMetrics)),
                   "bestIndex": int(self.bestIndex)},
        )
        self.bestModel.save(os.path.join(path, "bestModel"), overwrite=True)

    def _load_extra(self, path: str, meta: dict):
        import os

        from . import persistence

        self.avgMetrics = meta.get("avgMetrics", [])
        self.bestIndex = int(meta.get("bestIndex", 0))
        self.bestModel = persistence.load_instance(
            os.path.join(path, "bestModel")
        )


class TrainValidationSplit(CrossValidator):
    """Single split selection (Spark's TrainValidationSplit)."""

    def __init__(self, uid=None, trainRatio=0.75, **kw):
        super().__init__(uid, **kw)
        self.trainRatio = trainRatio

    def setTrainRatio(self, v):
        self.trainRatio = v
        return self

    def _fit(self, dataset: TensorFrame) -> CrossValidatorModel:
        assert self.estimator is not None and self.evaluator is not None
        g = torch.Generator(device="cpu").manual_seed(self.seed)
        mask = (torch.rand(dataset.count(), generator=g)
                < self.trainRatio).to(dataset.device)
        tr, va = dataset.filter(mask), dataset.filter(~mask)
        scores = []
        for pm in self.estimatorParamMaps:
            model = self.estimator.fit(tr, pm or None)
            scores.append(self.evaluator.evaluate(model.transform(va)))
        better = max if self.evaluator.isLargerBetter() else min
        best_idx = scores.index(better(scores))
        out = CrossValidatorModel()
        out.bestModel = self.estimator.fit(
            dataset, self.estimatorParamMaps[best_idx] or None
        )
        out.avgMetrics = scores
        out.bestIndex = best_idx
        return out
