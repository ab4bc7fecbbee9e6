"""Per-phase grower timing at config-5 member-fit scale (SEA_GROW_PROF)."""
import os
import sys
import time

import torch

os.environ["SEA_GROW_PROF"] = "1"
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.parallel import Comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_regression

set_comm(Comm())
df = synthetic_regression(1_600_000, 256, seed=9, device="cuda:0")
est = sea.GBMRegressor().setNumBaseLearners(5)
est.fit(df)  # warm
torch.cuda.synchronize()
t0 = time.time()
est.fit(df)
torch.cuda.synchronize()
print(f"gbm5 wall {(time.time()-t0)*1000:.1f} ms")
