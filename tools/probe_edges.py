"""Edge-shape robustness probe: wide F, small bins, weighted wide-K."""
import sys
import torch

sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.models import DecisionTreeClassifier, DecisionTreeRegressor
from spark_ensemble_amd.utils.io import synthetic_classification, synthetic_regression

# wide features
df = synthetic_regression(2_000_000, 2048, seed=3, device="cuda:0")
m = sea.GBMRegressor().setNumBaseLearners(3).setBaseLearner(
    DecisionTreeRegressor().setMaxDepth(6).setMaxBins(256)).fit(df)
torch.cuda.synchronize(); print("wide F=2048 ok", flush=True)

# small bins
df2 = synthetic_classification(5_000_000, 256, k=2, seed=4, device="cuda:0")
m2 = sea.GBMClassifier().setNumBaseLearners(3).setBaseLearner(
    DecisionTreeRegressor().setMaxDepth(8).setMaxBins(16)).fit(df2)
torch.cuda.synchronize(); print("bins=16 ok", flush=True)

# weighted wide multiclass (C > 8 chunked hist + eager split)
g = torch.Generator(device="cuda:0").manual_seed(5)
x = torch.randn(500_000, 32, generator=g, device="cuda:0")
y = torch.randint(0, 12, (500_000,), generator=g, device="cuda:0").float()
w = torch.rand(500_000, generator=g, device="cuda:0") + 0.5
fr = TensorFrame(features=x, label=y, weight=w)
mc = DecisionTreeClassifier().setMaxDepth(7).setMaxBins(128).setWeightCol(
    "weight").fit(fr)
torch.cuda.synchronize(); print("weighted K=12 gini ok", flush=True)

# single-row / tiny fits
tiny = TensorFrame(features=torch.randn(3, 4, device="cuda:0"),
                   label=torch.tensor([0., 1., 0.], device="cuda:0"))
sea.GBMClassifier().setNumBaseLearners(2).fit(tiny)
torch.cuda.synchronize(); print("tiny n=3 ok", flush=True)
print("EDGES OK", flush=True)
