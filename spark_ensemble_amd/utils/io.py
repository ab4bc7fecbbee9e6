"""Data loading: libsvm files and synthetic tabular generators.

The reference's test datasets are libsvm format (reference data/*.svm,
loaded at e.g. GBMClassifierSuite.scala:53-58); the bench uses synthetic
data of the BASELINE.json shapes (no network access for real datasets).
"""

from __future__ import annotations

from typing import Optional

import numpy as np
import torch

from ..frame import TensorFrame


def load_libsvm(path: str, num_features: Optional[int] = None, device=None) -> TensorFrame:
    """Dense-load a libsvm file: 'label idx:val idx:val ...' (1-based idx)."""
    labels = []
    rows = []
    cols = []
    vals = []
    with open(path) as f:
        for i, line in enumerate(f):
            parts = line.split()
            if not parts:
                continue
            labels.append(float(parts[0]))
            for tok in parts[1:]:
                j, v = tok.split(":")
                rows.append(i)
                cols.append(int(j) - 1)
                vals.append(float(v))
    n = len(labels)
    f_count = num_features or (max(cols) + 1 if cols else 0)
    x = np.zeros((n, f_count), dtype=np.float32)
    x[rows, cols] = vals
    y = np.asarray(labels, dtype=np.float32)
    # normalize +-1 labels to {0,1} for classification-style files
    uniq = np.unique(y)
    if set(uniq.tolist()) == {-1.0, 1.0}:
        y = (y + 1.0) / 2.0
    xt = torch.from_numpy(x)
    yt = torch.from_numpy(y)
    if device is not None:
        xt, yt = xt.to(device), yt.to(device)
    return TensorFrame(features=xt, label=yt)


def synthetic_classification(
    n: int,
    f: int,
    k: int = 2,
    seed: int = 7,
    device=None,
    informative: Optional[int] = None,
    shard: Optional[tuple] = None,
    noise: float = 0.35,
    split: int = 0,
) -> TensorFrame:
    """Gaussian + nonlinear-boundary synthetic classification data.

    ``seed`` fixes the labeling FUNCTION (a random linear map plus an
    interaction term); ``split``/``shard=(rank, world)`` select independent
    row draws from the same task, so train (split=0) and test (split=1)
    frames share a distribution.  Learnable by trees and by logistic
    regression, not trivially separable.

    NOTE: rows are drawn from a DEVICE-LOCAL generator (CPU
    and GPU torch RNG engines differ), so the same ``seed`` gives
    different rows on CPU vs GPU.  The distribution (and hence every
    statistical-quality assertion) is device-independent; only row-exact
    CPU-vs-GPU comparisons must generate on one device and ``.to()`` the
    other — which is what the parity tests in tests/test_gpu.py do.
    """
    rank, world = shard or (0, 1)
    task = torch.Generator().manual_seed(seed * 9176 + 4242)
    informative = min(informative or min(f, 32), f)
    wm = torch.randn(informative, k, generator=task)
    wi = torch.randn(1, k, generator=task)

    n_local = n // world + (1 if rank < n % world else 0)
    dev = torch.device(device) if device is not None else torch.device("cpu")
    # generate directly on the target device (bench-scale CPU randn of a
    # 10M x 256 shard costs ~10 s; on-GPU it is milliseconds)
    g = torch.Generator(device=dev).manual_seed(
        seed * 9176 + 100003 * (split + 1) + rank
    )
    x = torch.randn(n_local, f, generator=g, device=dev)
    wm_d, wi_d = wm.to(dev), wi.to(dev)
    margin = x[:, :informative] @ wm_d
    margin = margin + (x[:, 0] * x[:, 1]).unsqueeze(1) * wi_d
    margin = margin + noise * torch.randn(n_local, k, generator=g, device=dev)
    y = margin.argmax(dim=1).float()
    return TensorFrame(features=x, label=y)


def synthetic_regression(
    n: int,
    f: int,
    seed: int = 7,
    device=None,
    informative: Optional[int] = None,
    shard: Optional[tuple] = None,
    noise: float = 0.1,
    split: int = 0,
) -> TensorFrame:
    rank, world = shard or (0, 1)
    task = torch.Generator().manual_seed(seed * 7919 + 2424)
    informative = min(informative or min(f, 32), f)
    w = torch.randn(informative, generator=task)

    n_local = n // world + (1 if rank < n % world else 0)
    dev = torch.device(device) if device is not None else torch.device("cpu")
    g = torch.Generator(device=dev).manual_seed(
        seed * 7919 + 100003 * (split + 1) + rank
    )
    x = torch.randn(n_local, f, generator=g, device=dev)
    y = x[:, :informative] @ w.to(dev)
    y = y + 0.5 * (x[:, 0] * x[:, 1]) + torch.sin(x[:, 2] * 2.0)
    y = y + noise * torch.randn(n_local, generator=g, device=dev)
    return TensorFrame(features=x, label=y)


def from_pandas(df, features_cols=None, label_col="label", weight_col=None,
                device=None) -> TensorFrame:
    """Build a TensorFrame from a pandas DataFrame (the on-ramp for users
    coming from Spark DataFrames).  ``features_cols`` defaults to every
    numeric or categorical column except the label/weight.  pandas
    ``category`` columns are factorized to 0-based codes and recorded as
    categorical feature metadata (reference Utils.getFeaturesMetadata,
    Utils.scala:42-61), so tree splits land on exact category ids."""
    import pandas as pd

    def _usable(c):
        return isinstance(df[c].dtype, pd.CategoricalDtype) or np.issubdtype(
            df[c].dtype, np.number
        )

    if features_cols is None:
        features_cols = [
            c for c in df.columns
            if c not in (label_col, weight_col) and _usable(c)
        ]
    mats = []
    categorical = {}
    for j, c in enumerate(features_cols):
        if isinstance(df[c].dtype, pd.CategoricalDtype):
            codes = df[c].cat.codes.to_numpy()
            categorical[j] = len(df[c].cat.categories)
            mats.append(codes.astype(np.float32))
        else:
            mats.append(df[c].to_numpy(dtype=np.float32))
    x = torch.from_numpy(np.ascontiguousarray(np.stack(mats, axis=1)))
    cols = {"features": x}
    if label_col in df.columns:
        cols["label"] = torch.from_numpy(
            df[label_col].to_numpy(dtype=np.float32)
        )
    if weight_col and weight_col in df.columns:
        cols["weight"] = torch.from_numpy(
            df[weight_col].to_numpy(dtype=np.float32)
        )
    fr = TensorFrame(cols)
    if categorical:
        fr.set_categorical(categorical)
    return fr.to(device) if device is not None else fr


def read_parquet(path: str, features_cols=None, label_col="label",
                 weight_col=None, device=None) -> TensorFrame:
    """Load a parquet file into a TensorFrame (pyarrow)."""
    import pyarrow.parquet as pq

    table = pq.read_table(path)
    return from_pandas(
        table.to_pandas(), features_cols, label_col, weight_col, device
    )


def read_csv(path: str, features_cols=None, label_col="label",
             weight_col=None, device=None, **kw) -> TensorFrame:
    """Load a CSV file into a TensorFrame (pandas reader kwargs pass
    through)."""
    import pandas as pd

    return from_pandas(pd.read_csv(path, **kw), features_cols, label_col,
                       weight_col, device)
