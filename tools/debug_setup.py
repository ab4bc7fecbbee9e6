import sys, time
import torch
sys.path.insert(0, __import__("os").path.dirname(__import__("os").path.dirname(__import__("os").path.abspath(__file__))))
import spark_ensemble_amd as sea
from spark_ensemble_amd.ensemble.binning import BinnedDataset
from spark_ensemble_amd.ensemble.utils import subspace
from spark_ensemble_amd.boosting.losses import get_classification_loss
from spark_ensemble_amd.frame import TensorFrame
from spark_ensemble_amd.models.dummy import DummyClassifier
from spark_ensemble_amd.parallel import Comm, get_comm, set_comm
from spark_ensemble_amd.utils.io import synthetic_classification

set_comm(Comm())
df = synthetic_classification(10_000_000, 256, k=2, seed=1234,
                              device="cuda:0", informative=48)
est = sea.GBMClassifier()

def t(name, fn):
    torch.cuda.synchronize(); t0 = time.time()
    out = fn()
    torch.cuda.synchronize()
    print(f"{name:24s} {(time.time()-t0)*1000:8.1f} ms")
    return out

for rep in range(2):
    print(f"-- rep {rep} --")
    x, y, w = t("extract", lambda: est._extract_xyw(df))
    comm = get_comm()
    t("num_classes", lambda: int(comm.all_reduce_scalar(est._get_num_classes(df), "max")))
    binned = t("BinnedDataset", lambda: BinnedDataset(x, df))
    t("binned.get", lambda: binned.get(256))
    prior = t("prior fit", lambda: DummyClassifier().setStrategy("prior").fit(
        TensorFrame(features=x, label=y, weight=w)))
    loss = get_classification_loss("bernoulli", 2)
    t("encode", lambda: loss.encode_label(y))
    t("subspaces x100", lambda: [subspace(1.0, 256, 7 + i) for i in range(100)])
