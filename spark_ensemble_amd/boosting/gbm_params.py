"""GBM shared params (reference boosting/GBMParams.scala:26-136).

Defaults (reference :121-129): optimizedWeights=true, updates=gradient,
learningRate=1.0, numBaseLearners=10, tol=1e-6, maxIter=100, numRounds=1,
validationTol=0.01, and replacement=false for GBM's sub-bagging.
"""

from __future__ import annotations

from ..ensemble.params import (
    HasAggregationDepth,
    HasBaseLearner,
    HasCheckpointInterval,
    HasNumBaseLearners,
    HasSubBag,
)
from ..params import Params, ParamValidators

SUPPORTED_UPDATES = ["newton", "gradient"]


class GBMParams(
    HasNumBaseLearners,
    HasBaseLearner,
    HasSubBag,
    HasCheckpointInterval,
    HasAggregationDepth,
):
    def _declare_params(self):
        super()._declare_params()
        self.optimizedWeights = self._bool_param(
            "optimizedWeights", "line-search the stage weights"
        )
        self.updates = self._str_param(
            "updates",
            "pseudo-residual mode: newton or gradient",
            ParamValidators.inArray(SUPPORTED_UPDATES),
        )
        self.learningRate = self._float_param(
            "learningRate", "shrinkage per stage", ParamValidators.gt(0.0)
        )
        self.validationTol = self._float_param(
            "validationTol",
            "early-stop tolerance on validation loss",
            ParamValidators.gtEq(0.0),
        )
        self.numRounds = self._int_param(
            "numRounds",
            "patience rounds before early stop",
            ParamValidators.gtEq(1),
        )
        self.maxIter = self._int_param(
            "maxIter", "max line-search iterations", ParamValidators.gtEq(1)
        )
        self.tol = self._float_param(
            "tol", "line-search / quantile tolerance", ParamValidators.gtEq(0.0)
        )
        self.validationIndicatorCol = self._str_param(
            "validationIndicatorCol",
            "boolean column marking validation rows",
            lower=False,
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(
            optimizedWeights=True,
            updates="gradient",
            learningRate=1.0,
            numBaseLearners=10,
            tol=1e-6,
            maxIter=100,
            numRounds=1,
            validationTol=0.01,
            replacement=False,
            subsampleRatio=1.0,
            subspaceRatio=1.0,
            seed=0,
        )

    def setOptimizedWeights(self, v):
        return self.set("optimizedWeights", v)

    def setUpdates(self, v):
        return self.set("updates", v)

    def setLearningRate(self, v):
        return self.set("learningRate", v)

    def setValidationTol(self, v):
        return self.set("validationTol", v)

    def setNumRounds(self, v):
        return self.set("numRounds", v)

    def setMaxIter(self, v):
        return self.set("maxIter", v)

    def setTol(self, v):
        return self.set("tol", v)

    def setValidationIndicatorCol(self, v):
        return self.set("validationIndicatorCol", v)

    def setSeed(self, v):
        return self.set("seed", v)
