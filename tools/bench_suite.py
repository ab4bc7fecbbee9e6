"""Secondary benchmark suite: measured 1-GPU numbers for BASELINE.json
configs 3/4/5 (the driver's bench.py covers config 2). Prints one JSON
line per config."""
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.models import DecisionTreeRegressor, LinearRegression, LogisticRegression
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression

DEV = "cuda:0"


def timed(fn):
    torch.cuda.synchronize()
    t0 = time.time()
    out = fn()
    torch.cuda.synchronize()
    return out, time.time() - t0


def report(name, rows, work_units, secs, extra=None):
    print(json.dumps({
        "bench": name, "rows": rows, "work_units": work_units,
        "seconds": round(secs, 3),
        "rows_per_sec": round(rows * work_units / secs),
        **(extra or {}),
    }))


def config3():
    # BaggingRegressor, DecisionTree base, subspace 0.5 (10 of config's 100
    # estimators; rows/sec extrapolates per-estimator)
    n, f, k = 10_000_000, 256, 10
    df = synthetic_regression(n, f, seed=5, device=DEV)
    est = (sea.BaggingRegressor()
           .setBaseLearner(DecisionTreeRegressor().setMaxDepth(8).setMaxBins(256))
           .setNumBaseLearners(k).setSubspaceRatio(0.5).setSubsampleRatio(0.8)
           .setReplacement(True).setSeed(3))
    est.fit(df)  # warm (bins cached on frame after first; JIT warm)
    _, secs = timed(lambda: est.fit(df))
    report("config3_bagging_regressor_1gpu", n, k, secs,
           {"model": "BaggingRegressor(DT depth8, subspace=0.5, 10 est)"})


def config4():
    # BoostingClassifier SAMME.R with LogisticRegression base (fused GPU
    # loss/grad + L-BFGS); 1024 features exercises the wide-row path
    n, f, k = 5_000_000, 1024, 3
    df = synthetic_classification(n, f, k=2, seed=7, device=DEV, informative=64)
    est = (sea.BoostingClassifier()
           .setBaseLearner(LogisticRegression().setMaxIter(20))
           .setNumBaseLearners(k).setAlgorithm("real"))
    est.fit(df)
    _, secs = timed(lambda: est.fit(df))
    report("config4_boosting_logreg_1gpu", n, k, secs,
           {"model": "BoostingClassifier(SAMME.R, LogisticRegression lbfgs20, 3 rounds)",
            "features": f})


def config5():
    # StackingRegressor, GBM + Bagging bases, linear meta, 5-fold OOF
    n = 2_000_000
    df = synthetic_regression(n, 256, seed=9, device=DEV)
    est = (sea.StackingRegressor()
           .setBaseLearners([
               sea.GBMRegressor().setNumBaseLearners(5),
               sea.BaggingRegressor().setNumBaseLearners(5).setSubspaceRatio(0.7),
           ])
           .setStacker(LinearRegression())
           .setParallelism(4)
           .setNumFolds(5).setSeed(1))
    est.fit(df)  # warm (bins cache is per-fit, but kernels/allocator warm)
    _, secs = timed(lambda: est.fit(df))
    # work: 2 base learners x 5 folds + final refits + stacker
    report("config5_stacking_oof_1gpu", n, 1, secs,
           {"model": "StackingRegressor(GBM5+Bagging5, linear meta, 5-fold OOF)"})


if __name__ == "__main__":
    which = sys.argv[1:] or ["3", "4", "5"]
    for w in which:
        {"3": config3, "4": config4, "5": config5}[w]()
