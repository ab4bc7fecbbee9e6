"""Why does the bench's cold-frame public fit cost more per round than a
warm refit?  Times fit() on a FRESH TensorFrame (bins cache cold — the
bench contract) vs the same tensors in a warm frame, plus instr round
deltas for the cold case."""
import statistics
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.models import DecisionTreeRegressor
from spark_ensemble_amd.parallel import Comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_classification

set_comm(Comm())
df = synthetic_classification(10_000_000, 256, k=2, seed=1234,
                              device="cuda:0", informative=48)
K = 20


def mk_est(k=K):
    return (sea.GBMClassifier().setLoss("bernoulli").setNumBaseLearners(k)
            .setBaseLearner(DecisionTreeRegressor().setMaxDepth(8).setMaxBins(256))
            .setSeed(1234))


def run(fresh):
    frame = (TensorFrame(features=df["features"], label=df["label"])
             if fresh else df)
    est = mk_est()
    torch.cuda.synchronize()
    t0 = time.time()
    est.fit(frame)
    torch.cuda.synchronize()
    return time.time() - t0, est


# warm everything once (kernels, allocator, and df's own bins cache)
mk_est(5).fit(df)

for rep in range(3):
    t_cold, est_c = run(True)
    t_warm, _ = run(False)
    print(f"rep {rep}: cold {t_cold*1000:.0f}ms ({t_cold*1000/K:.1f}/round)  "
          f"warm {t_warm*1000:.0f}ms ({t_warm*1000/K:.1f}/round)  "
          f"delta {(t_cold-t_warm)*1000:.0f}ms")

h = est_c._instr.history
d = [h[i]["ms"] - h[i - 1]["ms"] for i in range(1, len(h))]
print("cold fit: h0", round(h[0]["ms"], 1), "setup_ms",
      round(est_c._instr.timers.get("setup_ms", -1), 1))
print("round deltas:", [round(v, 1) for v in d])
print("mean", round(statistics.mean(d), 2))
