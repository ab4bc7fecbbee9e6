"""300-round flagship soak: flat round times + stable VRAM after the
late-r02 grower/pipeline rework (pinned staging, events, caches)."""
import json
import sys
import time

import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.utils.io import synthetic_classification

df = synthetic_classification(2_000_000, 128, k=2, seed=3, device="cuda:0")
est = (sea.GBMClassifier().setLoss("bernoulli").setNumBaseLearners(300)
       .setUpdates("newton"))
t0 = time.time()
model = est.fit(df)
torch.cuda.synchronize()
el = time.time() - t0
mem = torch.cuda.max_memory_allocated() / (1 << 30)
rounds = len(model.models)
print(json.dumps({
    "soak": "gbm300_2Mx128", "rounds": rounds,
    "total_s": round(el, 2), "ms_per_round": round(1000 * el / rounds, 2),
    "max_mem_gib": round(mem, 2),
}))
# second fit on a fresh frame: steady-state check (caches, pinned pools)
df2 = synthetic_classification(2_000_000, 128, k=2, seed=4, device="cuda:0")
t0 = time.time()
est.fit(df2)
torch.cuda.synchronize()
print(json.dumps({"soak2_total_s": round(time.time() - t0, 2),
                  "max_mem_gib": round(torch.cuda.max_memory_allocated() / (1 << 30), 2)}))
