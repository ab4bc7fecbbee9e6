"""Find the first nondeterministic intermediate in a GPU GBM fit."""
import hashlib
import sys
import torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.boosting.losses import get_classification_loss
from spark_ensemble_amd.boosting.line_search import optimize_weight_1d
from spark_ensemble_amd.ensemble.binning import BinnedDataset
from spark_ensemble_amd.models import DecisionTreeRegressor
from spark_ensemble_amd.utils.io import synthetic_classification
from spark_ensemble_amd.ops import dispatch as ops

def h(t):
    return hashlib.md5(t.detach().cpu().numpy().tobytes()).hexdigest()[:10]

dev = "cuda:0"
df = synthetic_classification(100000, 24, k=2, seed=44, device=dev)
x, y = df["features"], df["label"]

def run():
    out = []
    loss = get_classification_loss("bernoulli", 2)
    learner = DecisionTreeRegressor().setMaxDepth(6).setMaxBins(256)
    binned = BinnedDataset(x, df)
    edges, bins = binned.get(256)
    out.append(("bins", h(bins)))
    ylab = loss.encode_label(y)
    import math
    p1 = float(y.mean()); logodds = math.log(p1/(1-p1))
    pred = torch.full((x.shape[0], 1), logodds, device=dev)
    gbm = sea.GBMClassifier()
    for r in range(3):
        g, _ = loss.grad_hess_fused(ylab, pred)
        out.append((f"r{r}.grad", h(g)))
        res = -g.squeeze(1)
        fr = binned.fit_frame(learner, res)
        model = gbm.fit_base_learner(learner, fr)
        out.append((f"r{r}.tree_feat", h(model._tree["feature"])))
        out.append((f"r{r}.tree_leaf", h(model._tree["leaf_value"])))
        d = getattr(model, "_train_pred", None)
        if d is None:
            d = model.predict(x)
        out.append((f"r{r}.dir", h(d)))
        w = torch.ones(x.shape[0], device=dev)
        a = optimize_weight_1d(loss, ylab, pred, d.unsqueeze(1), w, None)
        out.append((f"r{r}.alpha", f"{a:.17g}"))
        pred = pred + a * d.unsqueeze(1)
        out.append((f"r{r}.pred", h(pred)))
    return out

a1, a2 = run(), run()
bad = False
for (k1, v1), (k2, v2) in zip(a1, a2):
    mark = "" if v1 == v2 else "   <-- DIFFERS"
    if v1 != v2 and not bad:
        bad = True
    print(f"{k1:16s} {v1} {v2}{mark}")
print("deterministic" if not bad else "NONDETERMINISTIC")
