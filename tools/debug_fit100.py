import statistics
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.models import DecisionTreeRegressor
from spark_ensemble_amd.parallel import Comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_classification

K = int(sys.argv[1]) if len(sys.argv) > 1 else 40

set_comm(Comm())
df = synthetic_classification(10_000_000, 256, k=2, seed=1234,
                              device="cuda:0", informative=48)
est = (sea.GBMClassifier().setLoss("bernoulli").setNumBaseLearners(K)
       .setBaseLearner(DecisionTreeRegressor().setMaxDepth(8).setMaxBins(256)))
est.fit(df)
torch.cuda.synchronize()
t0 = time.time()
est.fit(df)
torch.cuda.synchronize()
wall = time.time() - t0
h = est._instr.history
d = [h[i]["ms"] - h[i - 1]["ms"] for i in range(1, len(h))]
print(f"K={K} wall {wall:.2f}s ({wall*1000/K:.1f} ms/round)  h0 {h[0]['ms']:.0f}ms  "
      f"rounds mean {statistics.mean(d):.1f} min {min(d):.1f} max {max(d):.1f}")
print("setup_ms", round(est._instr.timers.get("setup_ms", -1), 1))
print("every 10th:", [round(v, 1) for v in d[::10]])
print("last 10:", [round(v, 1) for v in d[-10:]])
ev = [r.get("ls_evals") for r in h]
print("ls_evals every 10th:", ev[::10])
