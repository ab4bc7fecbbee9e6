from .bagging import BaggingClassificationModel, BaggingClassifier  # noqa: F401
from .boosting import BoostingClassificationModel, BoostingClassifier  # noqa: F401
from .gbm import GBMClassificationModel, GBMClassifier  # noqa: F401
from .stacking import StackingClassificationModel, StackingClassifier  # noqa: F401
