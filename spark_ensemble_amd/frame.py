"""Columnar tensor frame — the MI355X-native replacement for Spark DataFrames.

The reference keeps rows in Spark ``DataFrame``/``RDD[Instance]`` and pays a
driver<->executor round trip for every action.  Here a
dataset is a dict of named torch tensors sharing the row dimension, resident
in HBM on one GPU per rank; multi-GPU datasets are row-sharded, one shard per
rank (one process per GPU over RCCL).  All per-row maps in the reference
(slice / error / residual / margin update, e.g. reference
BoostingRegressor.scala:231-232, GBMClassifier.scala:437-449) become tensor
ops or fused HIP kernels over these columns.

Conventions:
  * ``features`` column: float32 tensor [N, F]
  * ``label``: float32 [N] (class index as float for classification, like
    Spark ML), ``weight``: float32 [N]
  * prediction columns appended by ``Model.transform``: ``prediction`` [N],
    ``rawPrediction`` [N, K], ``probability`` [N, K]
"""

from __future__ import annotations

from typing import Dict, Optional

import torch


class TensorFrame:
    """A columnar, device-resident data frame (row-sharded across ranks)."""

    def __init__(self, columns: Optional[Dict[str, torch.Tensor]] = None, **kw):
        cols: Dict[str, torch.Tensor] = {}
        if columns:
            cols.update(columns)
        cols.update(kw)
        n = None
        for name, t in cols.items():
            if not isinstance(t, torch.Tensor):
                t = torch.as_tensor(t)
                cols[name] = t
            if n is None:
                n = t.shape[0]
            elif t.shape[0] != n:
                raise ValueError(
                    f"column {name!r} has {t.shape[0]} rows, expected {n}"
                )
        self._cols = cols
        # Shared derived-data cache (e.g. quantile-binned feature matrices),
        # keyed by the identity of the source tensor so row/column edits
        # can never alias stale entries.  Propagated (same dict object)
        # through column-level transforms.
        self._cache: Dict = {}
        # Per-feature metadata (reference Utils.getFeaturesMetadata,
        # Utils.scala:42-61 — AttributeGroup column metadata carrying
        # categorical info): {feature_index: cardinality}.  Categorical
        # features hold integer category ids 0..card-1 in the float
        # features tensor (Spark ML's indexed-category convention) and
        # are binned by IDENTITY instead of quantiles, so tree splits land
        # on exact category-id boundaries (ordinal-categorical handling;
        # propagation through subspace slices is free because the sliced
        # edge rows travel with the feature — ensemble/binning.py).
        self._categorical: Dict[int, int] = {}

    def _with_cache_of(self, other: "TensorFrame") -> "TensorFrame":
        self._cache = other._cache
        self._categorical = other._categorical
        return self

    # -- feature metadata -------------------------------------------------
    def set_categorical(self, categorical: Dict[int, int]) -> "TensorFrame":
        """Declare categorical features: {feature_index: num_categories}."""
        self._categorical = {int(k): int(v) for k, v in categorical.items()}
        return self

    @property
    def categorical(self) -> Dict[int, int]:
        return self._categorical

    def cache_get(self, kind: str, src: torch.Tensor, key):
        ent = self._cache.get(kind)
        if ent is not None:
            src_ref, k, val = ent
            if src_ref is src and k == key:
                return val
        return None

    def cache_put(self, kind: str, src: torch.Tensor, key, val):
        self._cache[kind] = (src, key, val)

    # -- basic accessors --------------------------------------------------
    @property
    def columns(self):
        return list(self._cols)

    def __contains__(self, name):
        return name in self._cols

    def __getitem__(self, name) -> torch.Tensor:
        return self._cols[name]

    def count(self) -> int:
        """Local (this shard's) row count."""
        for t in self._cols.values():
            return int(t.shape[0])
        return 0

    @property
    def device(self):
        for t in self._cols.values():
            return t.device
        return torch.device("cpu")

    def num_features(self, features_col: str = "features") -> int:
        return int(self._cols[features_col].shape[1])

    # -- functional transforms -------------------------------------------
    def withColumn(self, name: str, tensor: torch.Tensor) -> "TensorFrame":
        cols = dict(self._cols)
        cols[name] = tensor
        return TensorFrame(cols)._with_cache_of(self)

    def drop(self, *names: str) -> "TensorFrame":
        cols = {k: v for k, v in self._cols.items() if k not in names}
        return TensorFrame(cols)._with_cache_of(self)

    def select(self, *names: str) -> "TensorFrame":
        return TensorFrame({k: self._cols[k] for k in names})._with_cache_of(self)

    def withRenamed(self, old: str, new: str) -> "TensorFrame":
        cols = {}
        for k, v in self._cols.items():
            cols[new if k == old else k] = v
        return TensorFrame(cols)._with_cache_of(self)

    def filter(self, mask: torch.Tensor) -> "TensorFrame":
        """Row filter by boolean mask or index tensor."""
        if mask.dtype == torch.bool:
            idx = mask.nonzero(as_tuple=True)[0]
        else:
            idx = mask
        return TensorFrame({k: v[idx] for k, v in self._cols.items()})._with_cache_of(self)

    def random_split(self, frac: float, seed: int = 0):
        """(train, test) split: Bernoulli(frac) per row (Spark
        ``randomSplit`` analog), deterministic in ``seed``."""
        g = torch.Generator(device="cpu").manual_seed(seed)
        mask = (torch.rand(self.count(), generator=g) < frac).to(self.device)
        return self.filter(mask), self.filter(~mask)

    def to(self, device) -> "TensorFrame":
        return TensorFrame({k: v.to(device) for k, v in self._cols.items()})._with_cache_of(self)

    def clone(self) -> "TensorFrame":
        return TensorFrame(dict(self._cols))._with_cache_of(self)

    def __len__(self):
        return self.count()

    def __repr__(self):
        parts = ", ".join(
            f"{k}:{tuple(v.shape)}:{str(v.dtype).replace('torch.', '')}"
            for k, v in self._cols.items()
        )
        return f"TensorFrame({parts}, device={self.device})"
