"""Ensemble utilities.

``weighted_median`` reproduces reference ensemble/Utils.scala:26-40 (sort by
value, first index where cumulative weight >= half of total); ``subspace`` /
``slice_features`` reproduce the sub-bagging primitives of reference
HasSubBag.scala:73-84 (Bernoulli(ratio) filter over feature indices with a
per-learner seed; sorted indices; dense gather).
"""

from __future__ import annotations

import torch


def weighted_median(values: torch.Tensor, weights: torch.Tensor) -> torch.Tensor:
    """Row-wise weighted median.

    values, weights: [N, M] (M models per row) or [M].  Returns [N] (or
    scalar).  Rule (reference Utils.scala:26-40): sort by value, take the
    first value whose cumulative weight >= 0.5 * total weight.
    """
    single = values.dim() == 1
    if single:
        values = values.unsqueeze(0)
        weights = weights.unsqueeze(0).expand_as(values)
    if weights.dim() == 1:
        weights = weights.unsqueeze(0).expand_as(values)
    order = values.argsort(dim=1)
    v_sorted = values.gather(1, order)
    w_sorted = weights.gather(1, order)
    cum = w_sorted.cumsum(dim=1)
    half = 0.5 * w_sorted.sum(dim=1, keepdim=True)
    # first index with cumulative weight >= half
    idx = (cum >= half).float().argmax(dim=1, keepdim=True)
    out = v_sorted.gather(1, idx).squeeze(1)
    return out[0] if single else out


def subspace(ratio: float, num_features: int, seed: int) -> torch.Tensor:
    """Bernoulli(ratio) feature-index filter, deterministic in ``seed``.

    Mirrors reference HasSubBag.scala:73-79 (XORShiftRandom(seed) Bernoulli
    over 0..numFeatures-1; indices come out sorted).  Guarantees at least one
    feature.  Ratio 1 is the identity (tested property, reference
    HasSubBagSuite.scala:60-105).
    """
    if ratio >= 1.0:
        return torch.arange(num_features, dtype=torch.long)
    g = torch.Generator().manual_seed(int(seed) & 0x7FFFFFFFFFFFFFFF)
    mask = torch.rand(num_features, generator=g) < ratio
    idx = mask.nonzero(as_tuple=True)[0]
    if idx.numel() == 0:
        idx = torch.randint(0, num_features, (1,), generator=g)
    return idx.sort().values


def slice_features(x: torch.Tensor, indices: torch.Tensor) -> torch.Tensor:
    """Dense feature gather [N, F] -> [N, |indices|]
    (reference HasSubBag.scala:81-84)."""
    if indices.numel() == x.shape[1]:
        # identity subspace — avoid the gather
        if bool((indices == torch.arange(x.shape[1], device=indices.device)).all()):
            return x
    return x.index_select(1, indices.to(x.device))


def packed_forest_margin(x, models, weights, subspaces, num_features,
                         cache=None):
    """Batched ensemble inference fast path: when every stage model is a
    built-in regression tree, the whole ensemble is ONE forest_predict
    kernel call instead of a launch (plus a feature-slice copy) per
    stage.  Non-identity subspaces are handled by remapping each tree's
    split-feature ids back into the ORIGINAL feature space, so the packed
    forest walks the unsliced x.  Returns [N] margins or None when a
    stage is not a tree.  ``cache``: caller-owned dict holding the packed
    arena across transform calls (models are immutable post-fit)."""
    from ..models.tree import DecisionTreeRegressionModel
    from ..ops import dispatch as _ops

    if not models:
        return None
    trees = []
    for m, sub in zip(models, subspaces):
        if not isinstance(m, DecisionTreeRegressionModel):
            return None
        t = m._tree
        identity = sub is None or (
            sub.numel() == num_features
            and bool((sub.cpu() == torch.arange(num_features)).all())
        )
        if not identity:
            remapped = getattr(m, "_tree_orig_feats", None)
            if remapped is None:
                feat = t["feature"].long()
                sub_dev = sub.to(feat.device)
                remapped = torch.where(
                    feat >= 0, sub_dev[feat.clamp_min(0)], feat.long()
                ).to(torch.int32)
                m._tree_orig_feats = remapped
            t = dict(t, feature=remapped)
        trees.append(t)
    w = torch.tensor([float(v) for v in weights], dtype=torch.float32)
    return _ops.forest_predict(x, trees, w, cache=cache).squeeze(1)


def packed_forest_vote(x, models, subspaces, num_features, soft,
                       cache=None):
    """Bagging-classifier voting as ONE packed forest_predict call.

    Exactness by leaf transform (cached per model): soft voting sums
    per-row probability vectors — equal to summing leaf vectors
    NORMALIZED per node; hard voting sums one-hot argmax votes — equal to
    summing per-node ONE-HOT(argmax(leaf)) vectors.  Returns [N, K] vote
    sums or None when a member is not a built-in classification tree."""
    from ..models.tree import DecisionTreeClassificationModel
    from ..ops import dispatch as _ops

    if not models:
        return None
    trees = []
    attr = "_leaf_soft" if soft else "_leaf_hard"
    for m, sub in zip(models, subspaces):
        if not isinstance(m, DecisionTreeClassificationModel):
            return None
        t = m._tree
        lv = getattr(m, attr, None)
        if lv is None:
            leaf = t["leaf_value"]
            if soft:
                lv = leaf / leaf.sum(dim=1, keepdim=True).clamp_min(1e-12)
            else:
                k = leaf.shape[1]
                lv = torch.zeros_like(leaf)
                lv.scatter_(1, leaf.argmax(dim=1, keepdim=True), 1.0)
            setattr(m, attr, lv)
        identity = sub is None or (
            sub.numel() == num_features
            and bool((sub.cpu() == torch.arange(num_features)).all())
        )
        feat = t["feature"]
        if not identity:
            remapped = getattr(m, "_tree_orig_feats", None)
            if remapped is None:
                fl = feat.long()
                sub_dev = sub.to(fl.device)
                remapped = torch.where(
                    fl >= 0, sub_dev[fl.clamp_min(0)], fl
                ).to(torch.int32)
                m._tree_orig_feats = remapped
            feat = remapped
        trees.append(dict(t, feature=feat, leaf_value=lv))
    w = torch.ones(len(trees), dtype=torch.float32)
    return _ops.forest_predict(x, trees, w, cache=cache)


def ensemble_feature_importances(models, weights, subspaces, num_features):
    """Weighted, subspace-mapped aggregate of member featureImportances
    (normalized to sum 1; zeros when no member exposes importances)."""
    agg = torch.zeros(num_features, dtype=torch.float64)
    for m, w, sub in zip(models, weights, subspaces):
        fi = getattr(m, "featureImportances", None)
        if fi is None:
            continue
        fi = fi.double() * abs(float(w))
        if sub is None or fi.numel() == num_features:
            agg[: fi.numel()] += fi
        else:
            agg.index_add_(0, sub.cpu().long(), fi)
    t = float(agg.sum())
    return (agg / t if t > 0 else agg).float()
