"""Per-depth step timing + phase attribution for one GBM round."""
import sys, time
import torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.boosting.losses import get_classification_loss
from spark_ensemble_amd.boosting.line_search import optimize_weight_1d
from spark_ensemble_amd.ensemble.binning import BinnedDataset
from spark_ensemble_amd.models import DecisionTreeRegressor
from spark_ensemble_amd.utils.io import synthetic_classification
from spark_ensemble_amd.parallel import Comm, set_comm

comm = Comm(); set_comm(comm)
dev = "cuda:0"
df = synthetic_classification(10_000_000, 256, k=2, seed=1234, device=dev, informative=48)
x, y = df["features"], df["label"]
loss = get_classification_loss("bernoulli")
binned = BinnedDataset(x, df)
binned.get(256)
ylab = loss.encode_label(y)
import math
p1 = float(y.mean()); predictions = torch.full((x.shape[0],1), math.log(p1/(1-p1)), device=dev)
gbm = sea.GBMClassifier()

def one_round(learner, pred):
    res = -loss.grad_hess_fused(ylab, pred)[0].squeeze(1)
    fr = binned.fit_frame(learner, res)
    model = gbm.fit_base_learner(learner, fr)
    d = getattr(model, "_train_pred", None)
    if d is None: d = model.predict(x)
    d = d.unsqueeze(1)
    w = torch.ones(x.shape[0], device=dev)
    a = optimize_weight_1d(loss, ylab, pred, d, w, comm, 100, 1e-6)
    return pred + a * d

for depth in (1, 2, 4, 6, 8):
    learner = DecisionTreeRegressor().setMaxDepth(depth).setMaxBins(256)
    p = predictions.clone()
    p = one_round(learner, p); p = one_round(learner, p)
    torch.cuda.synchronize(); t0 = time.time()
    for _ in range(6):
        p = one_round(learner, p)
    torch.cuda.synchronize()
    print(f"depth={depth}  ms/round={(time.time()-t0)/6*1000:.2f}")

# phase timing at depth 8 (syncs between phases — upper bounds)
learner = DecisionTreeRegressor().setMaxDepth(8).setMaxBins(256)
p = predictions.clone(); p = one_round(learner, p)
def t_phase(fn, reps=5):
    torch.cuda.synchronize(); t0=time.time()
    for _ in range(reps): out = fn()
    torch.cuda.synchronize(); return (time.time()-t0)/reps*1000, out
ms, res = t_phase(lambda: (-loss.grad_hess_fused(ylab, p)[0].squeeze(1)))
print(f"grad: {ms:.2f} ms")
fr = binned.fit_frame(learner, res)
ms, model = t_phase(lambda: gbm.fit_base_learner(learner, fr), 5)
print(f"tree fit: {ms:.2f} ms")
d = model._train_pred.unsqueeze(1)
w = torch.ones(x.shape[0], device=dev)
ms, a = t_phase(lambda: optimize_weight_1d(loss, ylab, p, d, w, comm, 100, 1e-6))
print(f"line search: {ms:.2f} ms")
ms, _ = t_phase(lambda: p + a * d)
print(f"margin: {ms:.2f} ms")
