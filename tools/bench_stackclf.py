"""Classification OOF stacking benchmark: fold-fused GBMClassifier bases
vs the sequential path (fold fusion disabled)."""
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
import spark_ensemble_amd.classification.gbm as gbm_mod
from spark_ensemble_amd.models import LogisticRegression
from spark_ensemble_amd.parallel import Comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_classification

set_comm(Comm())
DEV = "cuda:0"


def run(fused: bool):
    est = (
        sea.StackingClassifier()
        .setBaseLearners([
            sea.GBMClassifier().setNumBaseLearners(5),
            sea.GBMClassifier().setLoss("bernoulli").setNumBaseLearners(5),
        ])
        .setStacker(LogisticRegression().setMaxIter(10))
        .setStackMethod("proba")
        .setParallelism(2)
        .setNumFolds(5)
        .setSeed(1)
    )
    orig = gbm_mod.GBMClassifier._can_fit_folds
    if not fused:
        gbm_mod.GBMClassifier._can_fit_folds = lambda self: False
    try:
        est.fit(df)  # warm
        torch.cuda.synchronize()
        t0 = time.time()
        est.fit(df)
        torch.cuda.synchronize()
        return time.time() - t0
    finally:
        gbm_mod.GBMClassifier._can_fit_folds = orig


n = 2_000_000
# bernoulli needs binary labels
df = synthetic_classification(n, 256, k=2, seed=7, device=DEV)
t_fused = run(True)
t_seq = run(False)
print(json.dumps({
    "bench": "stacking_classifier_oof_2M_gbm_bases",
    "fused_s": round(t_fused, 3),
    "sequential_s": round(t_seq, 3),
    "rows_per_sec_fused": round(n / t_fused),
    "speedup": round(t_seq / t_fused, 2),
}))
