"""Multiclass GBM round cost: fused K-tree rounds vs sequential
per-class fits, at a letter-like 26-class shape scaled to GPU-relevant
row counts.  Reports ms/round and ms/tree for both paths."""
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
import spark_ensemble_amd.classification.gbm as gbm_mod
from spark_ensemble_amd.models import DecisionTreeRegressor
from spark_ensemble_amd.parallel import Comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_classification

set_comm(Comm())
DEV = "cuda:0"


def timed_fit(df, k_classes, rounds, fused):
    est = (sea.GBMClassifier().setLoss("logloss").setNumBaseLearners(rounds)
           .setBaseLearner(DecisionTreeRegressor().setMaxDepth(6).setMaxBins(256))
           .setSeed(3))
    orig = gbm_mod.GBMClassifier._can_fuse_round
    if not fused:
        gbm_mod.GBMClassifier._can_fuse_round = lambda self, l, w, st: False
    try:
        est.fit(df)  # warm
        torch.cuda.synchronize()
        t0 = time.time()
        est.fit(df)
        torch.cuda.synchronize()
        return (time.time() - t0) / rounds * 1000
    finally:
        gbm_mod.GBMClassifier._can_fuse_round = orig


def main():
    out = []
    for n, f, k, rounds in ((1_000_000, 64, 26, 3), (4_000_000, 64, 8, 3)):
        df = synthetic_classification(n, f, k=k, seed=6, device=DEV)
        ms_f = timed_fit(df, k, rounds, True)
        ms_s = timed_fit(df, k, rounds, False)
        out.append({
            "rows": n, "features": f, "classes": k,
            "fused_ms_per_round": round(ms_f, 2),
            "fused_ms_per_tree": round(ms_f / k, 3),
            "seq_ms_per_round": round(ms_s, 2),
            "speedup": round(ms_s / ms_f, 2),
        })
    print(json.dumps(out))


if __name__ == "__main__":
    main()
