"""Config-5 (OOF stacking) drilldown: where do the 0.5s go?
Times each member fit sequentially, then the whole stacking fit at
parallelism 1/2/4/8."""
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.models import DecisionTreeRegressor, LinearRegression
from spark_ensemble_amd.parallel import Comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_regression

set_comm(Comm())
DEV = "cuda:0"
n = 2_000_000
df = synthetic_regression(n, 256, seed=9, device=DEV)
x, y = df["features"], df["label"]


def t(name, fn, reps=1):
    torch.cuda.synchronize()
    t0 = time.time()
    for _ in range(reps):
        out = fn()
    torch.cuda.synchronize()
    dt = (time.time() - t0) / reps
    print(f"{name:42s} {dt*1000:8.1f} ms")
    return out


def gbm():
    return sea.GBMRegressor().setNumBaseLearners(5)


def bag():
    return sea.BaggingRegressor().setNumBaseLearners(5).setSubspaceRatio(0.7)


# warm
gbm().fit(df)
torch.cuda.synchronize()

shared = TensorFrame(features=x, label=y, weight=torch.ones_like(y))
t("gbm5 fit (shared frame, cold cache)", lambda: gbm().fit(shared))
t("gbm5 fit (warm cache)", lambda: gbm().fit(shared), reps=3)
t("bagging5 fit (warm cache)", lambda: bag().fit(shared), reps=3)
w0 = torch.ones_like(y)
w0[::5] = 0.0
t("gbm5 fold fit (20% zero weights)",
  lambda: gbm().fit(shared.withColumn("weight", w0)), reps=3)

for par in (1, 2, 4, 8):
    est = (sea.StackingRegressor()
           .setBaseLearners([gbm(), bag()])
           .setStacker(LinearRegression())
           .setParallelism(par)
           .setNumFolds(5).setSeed(1))
    t(f"stacking OOF fit parallelism={par}", lambda: est.fit(df), reps=2)
