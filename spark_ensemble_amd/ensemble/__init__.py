from .params import (  # noqa: F401
    HasAggregationDepth,
    HasBaseLearner,
    HasBaseLearners,
    HasCheckpointInterval,
    HasNumBaseLearners,
    HasParallelism,
    HasStacker,
    HasSubBag,
)
from .utils import weighted_median  # noqa: F401
