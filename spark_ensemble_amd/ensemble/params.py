"""Shared ensemble param traits.

Re-creates the ``private[ml]`` traits of reference
ensemble/ensembleParams.scala and ensemble/HasSubBag.scala:

  * HasNumBaseLearners        (ensembleParams.scala:32-49)
  * HasBaseLearner + the single choke-point ``fit_base_learner``
                              (ensembleParams.scala:51-83)
  * HasBaseLearners           (ensembleParams.scala:148-194)
  * HasStacker                (ensembleParams.scala:107-146)
  * HasSubBag                 (HasSubBag.scala:33-84)
  * HasParallelism / HasCheckpointInterval / HasAggregationDepth — Spark
    shared params used by BaggingParams/BoostingParams.

Persistence of nested estimators follows the MLlib subdirectory layout:
``learner/``, ``stacker/``, ``learner-<i>/`` (ensembleParams.scala:87-103,
118-133, 160-183).
"""

from __future__ import annotations

import os
from typing import List, Optional

import torch

from .. import persistence
from ..frame import TensorFrame
from ..params import Params, ParamValidators
from .utils import slice_features, subspace


class HasNumBaseLearners(Params):
    def _declare_params(self):
        super()._declare_params()
        self.numBaseLearners = self._int_param(
            "numBaseLearners",
            "number of base learners to fit",
            ParamValidators.gtEq(1),
        )

    def getNumBaseLearners(self):
        return self.getOrDefault("numBaseLearners")

    def setNumBaseLearners(self, v):
        return self.set("numBaseLearners", v)


class FitsBaseLearners(Params):
    # The single choke-point through which every meta-estimator fits its
    # base learner (reference ensembleParams.scala:64-81): re-target the
    # label/features/prediction/weight columns of a COPY of the learner and
    # call fit.
    def fit_base_learner(
        self,
        learner,
        dataset: TensorFrame,
        label_col: str = "label",
        features_col: str = "features",
        prediction_col: str = "prediction",
        weight_col: Optional[str] = None,
    ):
        lr = learner.copy()
        lr.set("labelCol", label_col)
        lr.set("featuresCol", features_col)
        lr.set("predictionCol", prediction_col)
        if weight_col is not None and lr.hasParam("weightCol"):
            lr.set("weightCol", weight_col)
        return lr.fit(dataset)


class HasBaseLearner(FitsBaseLearners):
    def _declare_params(self):
        super()._declare_params()
        self.baseLearner = self._param("baseLearner", "base learner estimator")

    def getBaseLearner(self):
        return self.getOrDefault("baseLearner")

    def setBaseLearner(self, v):
        return self.set("baseLearner", v)

    # -- persistence (learner/ subdir; ensembleParams.scala:87-103) --------
    def _save_learner(self, path: str):
        # baseLearner may be unset (the built-in tree default applies);
        # estimator saves then simply omit the learner/ subdir
        lr = self.getOrNone("baseLearner")
        if lr is not None:
            lr.save(os.path.join(path, "learner"), overwrite=True)

    @staticmethod
    def _load_learner(path: str):
        p = os.path.join(path, "learner")
        return persistence.load_instance(p) if os.path.isdir(p) else None


class HasBaseLearners(FitsBaseLearners):
    def _declare_params(self):
        super()._declare_params()
        self.baseLearners = self._param(
            "baseLearners", "list of base learner estimators"
        )

    def getBaseLearners(self) -> List:
        return self.getOrDefault("baseLearners")

    def setBaseLearners(self, v):
        return self.set("baseLearners", list(v))

    def _save_learners(self, path: str):
        for i, lr in enumerate(self.getBaseLearners()):
            lr.save(os.path.join(path, f"learner-{i}"), overwrite=True)

    @staticmethod
    def _load_learners(path: str) -> List:
        out = []
        i = 0
        while os.path.isdir(os.path.join(path, f"learner-{i}")):
            out.append(persistence.load_instance(os.path.join(path, f"learner-{i}")))
            i += 1
        return out


class HasStacker(FitsBaseLearners):
    def _declare_params(self):
        super()._declare_params()
        self.stacker = self._param("stacker", "meta-learner fit on base outputs")

    def getStacker(self):
        return self.getOrDefault("stacker")

    def setStacker(self, v):
        return self.set("stacker", v)

    def _save_stacker(self, path: str):
        self.getStacker().save(os.path.join(path, "stacker"), overwrite=True)

    @staticmethod
    def _load_stacker(path: str):
        return persistence.load_instance(os.path.join(path, "stacker"))


class HasParallelism(Params):
    """Driver-side task parallelism across independent base-learner fits
    (reference: Spark HasParallelism + ThreadUtils, used at
    BaggingRegressor.scala:145-166).  On MI355X independent fits on ONE GPU
    run on separate HIP streams; the param caps how many are in flight."""

    def _declare_params(self):
        super()._declare_params()
        self.parallelism = self._int_param(
            "parallelism",
            "max number of concurrent base-learner fits",
            ParamValidators.gtEq(1),
        )
        self._setDefault(parallelism=1)

    def getParallelism(self):
        return self.getOrDefault("parallelism")

    def setParallelism(self, v):
        return self.set("parallelism", v)


class HasCheckpointInterval(Params):
    def _declare_params(self):
        super()._declare_params()
        self.checkpointInterval = self._int_param(
            "checkpointInterval",
            "rounds between state snapshots (-1 disables)",
            lambda v: v == -1 or v >= 1,
        )
        self._setDefault(checkpointInterval=10)
        self._declare_checkpoint_dir()

    def getCheckpointInterval(self):
        return self.getOrDefault("checkpointInterval")

    def setCheckpointInterval(self, v):
        return self.set("checkpointInterval", v)

    # durable round-state dumps (SURVEY.md §5.4 rebuild of the reference's
    # PeriodicRDDCheckpointer — here fit() CAN resume a half-trained
    # ensemble from the dump; the reference's cannot)
    def getCheckpointDir(self):
        return self.getOrNone("checkpointDir")

    def setCheckpointDir(self, v):
        return self.set("checkpointDir", v)

    def _declare_checkpoint_dir(self):
        self.checkpointDir = self._str_param(
            "checkpointDir",
            "directory for durable round-state snapshots (resume on refit)",
            lower=False,
        )


class HasAggregationDepth(Params):
    """Kept for API parity with reference BoostingParams.scala:24,32; the
    RCCL all-reduce tree shape is chosen by the library, so this is advisory."""

    def _declare_params(self):
        super()._declare_params()
        self.aggregationDepth = self._int_param(
            "aggregationDepth",
            "suggested depth for tree aggregation (advisory under RCCL)",
            ParamValidators.gtEq(2),
        )
        self._setDefault(aggregationDepth=2)

    def getAggregationDepth(self):
        return self.getOrDefault("aggregationDepth")

    def setAggregationDepth(self, v):
        return self.set("aggregationDepth", v)


class HasSubBag(Params):
    """Row resampling + feature subspace params (reference HasSubBag.scala)."""

    def _declare_params(self):
        super()._declare_params()
        self.replacement = self._bool_param(
            "replacement", "sample rows with replacement"
        )
        self.subsampleRatio = self._float_param(
            "subsampleRatio",
            "fraction of rows sampled per learner",
            ParamValidators.inRange(0.0, 1.0, lower_inclusive=False),
        )
        self.subspaceRatio = self._float_param(
            "subspaceRatio",
            "fraction of features sampled per learner",
            ParamValidators.inRange(0.0, 1.0, lower_inclusive=False),
        )
        self._setDefault(replacement=True, subsampleRatio=1.0, subspaceRatio=1.0)

    def getReplacement(self):
        return self.getOrDefault("replacement")

    def setReplacement(self, v):
        return self.set("replacement", v)

    def getSubsampleRatio(self):
        return self.getOrDefault("subsampleRatio")

    def setSubsampleRatio(self, v):
        return self.set("subsampleRatio", v)

    def getSubspaceRatio(self):
        return self.getOrDefault("subspaceRatio")

    def setSubspaceRatio(self, v):
        return self.set("subspaceRatio", v)

    # -- primitives -------------------------------------------------------
    @staticmethod
    def subspace(ratio: float, num_features: int, seed: int) -> torch.Tensor:
        return subspace(ratio, num_features, seed)

    @staticmethod
    def slice_features(x: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
        return slice_features(x, indices)

    @staticmethod
    def sample_weights(
        replacement: bool,
        ratio: float,
        n: int,
        seed: int,
        device=None,
        base_weight: Optional[torch.Tensor] = None,
        rank: int = 0,
    ) -> torch.Tensor:
        """Row-resampling as a WEIGHT vector, the GPU-native form of
        ``RDD.sample`` (reference BaggingRegressor.scala:149-150,
        GBMRegressor.scala:357-359).

        replacement=True  -> Poisson(ratio) multiplicities (Spark's
                             PoissonSampler semantics for fraction<=1 with
                             replacement),
        replacement=False -> Bernoulli(ratio) 0/1 mask.

        Returning weights instead of materializing a resampled copy keeps
        the 288 GB-resident feature tensor un-moved; every downstream kernel
        (histograms, losses, errors) is weight-aware.  ``rank`` decorrelates
        shards so each GPU draws an independent stream.
        """
        from ..ops import dispatch as ops

        return ops.sample_weights(
            replacement, ratio, n, int(seed), device, base_weight, rank
        )
