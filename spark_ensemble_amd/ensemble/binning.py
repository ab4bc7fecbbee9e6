"""Shared binned-dataset helper for meta-estimators.

Every meta-estimator fits many base learners over the SAME resident feature
tensor (per GBM round, per bagged learner).  Tree base learners consume
quantile-binned uint8 features; binning costs a sort per feature, so it must
happen ONCE per meta-fit, not once per base fit.  ``BinnedDataset`` owns the
(edges, bins) pair per maxBins value and hands out fit frames whose cache is
pre-populated — including sliced copies for feature subspaces (reference
``slice`` at HasSubBag.scala:81-84 becomes a uint8 column gather here).
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from ..frame import TensorFrame
from ..ops import dispatch as ops
from .utils import slice_features


def apply_categorical_edges_(edges: torch.Tensor, categorical) -> torch.Tensor:
    """Overwrite quantile cut points with IDENTITY cut points for declared
    categorical features (reference Utils.getFeaturesMetadata semantics,
    Utils.scala:42-61: base learners must see categorical info — here the
    bin id IS the category id, so every split threshold is an exact
    category boundary).  edges[f] = [0, 1, ..., card-2, +BIG...]; the bin
    rule 'first edge >= v' then maps category k -> bin k.  Unseen ids
    > card-1 fall in the top bin."""
    if not categorical:
        return edges
    B1 = edges.shape[1]
    for f, card in categorical.items():
        if card - 1 > B1:
            raise ValueError(
                f"categorical feature {f} has {card} categories but maxBins "
                f"allows only {B1 + 1}; raise maxBins"
            )
        row = torch.full((B1,), 3.0e38, dtype=edges.dtype, device=edges.device)
        if card > 1:
            row[: card - 1] = torch.arange(
                card - 1, dtype=edges.dtype, device=edges.device
            )
        edges[f] = row
    return edges


def _is_identity(indices: Optional[torch.Tensor], num_features: int) -> bool:
    if indices is None:
        return True
    if indices.numel() != num_features:
        return False
    return bool(
        (indices.cpu() == torch.arange(num_features)).all()
    )


class BinnedDataset:
    def __init__(self, x: torch.Tensor, source_frame: Optional[TensorFrame] = None):
        self.x = x
        self._frame = source_frame
        self._by_bins: Dict[int, Tuple[torch.Tensor, torch.Tensor]] = {}
        # seed from any bins already cached on the source frame
        if source_frame is not None:
            for mb in (256, 128, 64, 32):
                hit = source_frame.cache_get("bins", x, mb)
                if hit is not None:
                    self._by_bins[mb] = hit

    def get(self, max_bins: int) -> Tuple[torch.Tensor, torch.Tensor]:
        if max_bins not in self._by_bins:
            edges = ops.quantile_bins(self.x, max_bins)
            if self._frame is not None:
                apply_categorical_edges_(edges, self._frame.categorical)
            from ..parallel import get_comm

            comm = get_comm()
            if comm.is_distributed:
                # identical cut points on every rank (see tree_grower)
                comm.broadcast_(edges, src=0)
            bins = ops.bin_features(self.x, edges)
            self._by_bins[max_bins] = (edges, bins)
            # write back to the source frame so REPEATED fits over the same
            # resident features (CV folds excluded — different tensors) skip
            # the ~2 s binning; without this every fresh fit re-paid it in
            # its first round
            if self._frame is not None:
                self._frame.cache_put("bins", self.x, max_bins,
                                      self._by_bins[max_bins])
        return self._by_bins[max_bins]

    def sliced_features(self, indices: Optional[torch.Tensor]) -> torch.Tensor:
        if _is_identity(indices, self.x.shape[1]):
            return self.x
        return slice_features(self.x, indices)

    def sliced_binned(self, indices: Optional[torch.Tensor], max_bins: int):
        """(edges, bins) for a feature subspace — the pre-binned inputs a
        fused forest fit consumes directly (fit_tree_forest)."""
        edges, bins = self.get(max_bins)
        if _is_identity(indices, self.x.shape[1]):
            return edges, bins
        idx_dev = indices.to(bins.device)
        return (
            edges.index_select(0, idx_dev).contiguous(),
            bins.index_select(1, idx_dev).contiguous(),
        )

    def fit_frame(
        self,
        learner,
        label: torch.Tensor,
        weight: Optional[torch.Tensor] = None,
        indices: Optional[torch.Tensor] = None,
        xs: Optional[torch.Tensor] = None,
    ) -> TensorFrame:
        """Build a {features, label, weight} frame for a base-learner fit,
        with pre-binned features in its cache when the learner is binned
        (has a maxBins param)."""
        identity = _is_identity(indices, self.x.shape[1])
        if xs is None:
            xs = self.x if identity else slice_features(self.x, indices)
        cols = {"features": xs, "label": label}
        if weight is not None:
            cols["weight"] = weight
        fr = TensorFrame(cols)
        if learner is not None and learner.hasParam("maxBins"):
            mb = int(learner.getOrDefault("maxBins"))
            edges, bins = self.get(mb)
            if not identity:
                idx_dev = indices.to(bins.device)
                edges = edges.index_select(0, idx_dev).contiguous()
                bins = bins.index_select(1, idx_dev).contiguous()
            fr.cache_put("bins", xs, mb, (edges, bins))
        return fr
