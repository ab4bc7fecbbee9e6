"""Inference (serving) benchmark: batched ensemble transform throughput.

Measures model.predict / predictRaw rows/sec for a trained 100-stage GBM
and a 50-tree bagging ensemble on 10M x 256 — the packed single-kernel
forest path (ensemble/utils.packed_forest_margin).
"""
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.models import DecisionTreeRegressor
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression

DEV = "cuda:0"


def timeit(fn, reps=5):
    fn(); torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.time() - t0) / reps


def both_modes(name, n, fn):
    import os

    t_b = timeit(fn)  # binned rank-transform mode (default when available)
    os.environ["SEA_SERVE_RAW"] = "1"
    t_r = timeit(fn)
    del os.environ["SEA_SERVE_RAW"]
    t = min(t_b, t_r)
    print(json.dumps({"bench": name, "rows": n,
                      "ms": round(t * 1000, 2),
                      "ms_binned": round(t_b * 1000, 2),
                      "ms_raw": round(t_r * 1000, 2),
                      "rows_per_sec": round(n / t)}))


def main():
    n, f = 10_000_000, 256
    dfc = synthetic_classification(n, f, k=2, seed=3, device=DEV, informative=48)
    gbm = (sea.GBMClassifier().setLoss("bernoulli").setNumBaseLearners(100)
           .setBaseLearner(DecisionTreeRegressor().setMaxDepth(8).setMaxBins(256))
           .fit(dfc))
    x = dfc["features"]
    both_modes("infer_gbm100_depth8", n, lambda: gbm.predictRaw(x))

    dfr = synthetic_regression(n, f, seed=4, device=DEV)
    bag = (sea.BaggingRegressor().setNumBaseLearners(50).setSubspaceRatio(0.5)
           .setBaseLearner(DecisionTreeRegressor().setMaxDepth(8).setMaxBins(256))
           .fit(dfr))
    both_modes("infer_bagging50_subspace", n, lambda: bag.predict(x))


if __name__ == "__main__":
    main()
