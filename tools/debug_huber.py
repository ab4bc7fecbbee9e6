import sys, torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.utils.io import synthetic_regression
from spark_ensemble_amd.boosting.losses import HuberLoss
from spark_ensemble_amd.boosting.line_search import _eval, optimize_weight_1d
from spark_ensemble_amd.ops import reference as ref

dev = "cuda:0"
g = torch.Generator().manual_seed(3)
n = 30000
y = torch.randn(n, 1, generator=g)
pred = torch.randn(n, 1, generator=g) * 0.3
d = (y - pred) * 0.6
w = torch.ones(n)
loss = HuberLoss(1.2)

for a in (0.0, 0.5, 1.0, 2.0):
    pc = _eval(loss, y, pred, d, w, a)
    pg = _eval(loss, y.to(dev), pred.to(dev), d.to(dev), w.to(dev), a).cpu()
    print(f"a={a}: cpu {pc.tolist()} gpu {pg.tolist()}")

ac = optimize_weight_1d(loss, y, pred, d, w, None)
ag = optimize_weight_1d(loss, y.to(dev), pred.to(dev), d.to(dev), w.to(dev), None)
print("alpha cpu", ac, "gpu", ag)

# grad_hess comparison
gc, _ = loss.grad_hess_fused(y, pred)
gg, _ = loss.grad_hess_fused(y.to(dev), pred.to(dev))
print("grad close:", torch.allclose(gc, gg.cpu(), atol=1e-5))

# full fit on GPU with instrumentation
df = synthetic_regression(50000, 32, seed=31, device=dev)
dft = synthetic_regression(20000, 32, seed=31, split=1, device=dev)
est = sea.GBMRegressor().setLoss("huber").setNumBaseLearners(5)
m = est.fit(df)
print("history:", est._instr.history)
p = m.predict(dft["features"])
ss = float(((p - dft["label"]) ** 2).mean())
print("ss", ss, "var", float(dft["label"].var()))
print("weights", m._weights)
