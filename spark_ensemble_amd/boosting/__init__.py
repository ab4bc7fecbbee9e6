from .losses import (  # noqa: F401
    AbsoluteLoss,
    BernoulliLoss,
    ExponentialLoss,
    GBMLoss,
    HuberLoss,
    LogCoshLoss,
    LogLoss,
    QuantileLoss,
    ScaledLogCoshLoss,
    SquaredLoss,
    get_regression_loss,
    get_classification_loss,
)
