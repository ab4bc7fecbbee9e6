from .stats import dist_mean, dist_quantile, dist_weighted_mean  # noqa: F401
