from .bagging import BaggingRegressionModel, BaggingRegressor  # noqa: F401
from .boosting import BoostingRegressionModel, BoostingRegressor  # noqa: F401
from .gbm import GBMRegressionModel, GBMRegressor  # noqa: F401
from .stacking import StackingRegressionModel, StackingRegressor  # noqa: F401
