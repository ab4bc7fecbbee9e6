"""Decision-tree base learners (the default base learners of every
meta-estimator, standing in for MLlib's DecisionTreeClassifier/Regressor
that the reference's tests use, e.g. reference BaggingClassifierSuite
.scala:48-78).

Param names/defaults follow Spark ML DecisionTree: maxDepth=5, maxBins=32,
minInstancesPerNode=1, minInfoGain=0.0, seed.

Both classes run on the unified histogram trainer (tree_grower.py):
  * regressor: D=1 targets  (variance-reduction splits, leaf = mean)
  * classifier: D=K one-hot targets (identically gini splits, leaf = class
    probability vector)
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from .. import persistence
from ..estimator import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    RegressionModel,
    Regressor,
)
from ..frame import TensorFrame
from ..ops import dispatch as ops
from ..params import Params, ParamValidators
from ..parallel import get_comm
from .tree_grower import GrowParams, ensure_binned, grow_tree


class _TreeParams(Params):
    def _declare_params(self):
        super()._declare_params()
        self.maxDepth = self._int_param(
            "maxDepth", "maximum tree depth", ParamValidators.inRange(0, 64)
        )
        self.maxBins = self._int_param(
            "maxBins", "number of histogram bins", ParamValidators.inRange(2, 256)
        )
        self.minInstancesPerNode = self._int_param(
            "minInstancesPerNode",
            "minimum rows per child",
            ParamValidators.gtEq(1),
        )
        self.minInfoGain = self._float_param(
            "minInfoGain", "minimum gain to split", ParamValidators.gtEq(0.0)
        )
        self.minWeightFractionPerNode = self._float_param(
            "minWeightFractionPerNode",
            "minimum fraction of total weight per child",
            ParamValidators.inRange(0.0, 0.5, upper_inclusive=False),
        )
        self.regLambda = self._float_param(
            "regLambda", "L2 regularization on leaf values", ParamValidators.gtEq(0.0)
        )
        self.seed = self._int_param("seed", "random seed")
        self._setDefault(
            maxDepth=5,
            maxBins=32,
            minInstancesPerNode=1,
            minInfoGain=0.0,
            minWeightFractionPerNode=0.0,
            regLambda=1e-6,
            seed=0,
        )

    def setMaxDepth(self, v):
        return self.set("maxDepth", v)

    def setMaxBins(self, v):
        return self.set("maxBins", v)

    def setMinInstancesPerNode(self, v):
        return self.set("minInstancesPerNode", v)

    def setMinInfoGain(self, v):
        return self.set("minInfoGain", v)

    def setSeed(self, v):
        return self.set("seed", v)

    def _grow_params(self, total_weight: float) -> GrowParams:
        return GrowParams(
            max_depth=self.getOrDefault("maxDepth"),
            max_bins=self.getOrDefault("maxBins"),
            min_instances_per_node=self.getOrDefault("minInstancesPerNode"),
            min_info_gain=self.getOrDefault("minInfoGain"),
            min_child_weight=self.getOrDefault("minWeightFractionPerNode")
            * total_weight,
            lam=self.getOrDefault("regLambda"),
        )


class _TreeModelMixin:
    """Shared storage/persistence for fitted tree models."""

    @property
    def numNodes(self) -> int:
        return int(self._tree["feature"].numel())

    @property
    def toDebugString(self) -> str:
        """Human-readable tree dump (MLlib ``toDebugString`` analog)."""
        t = {k: v.cpu() for k, v in self._tree.items()}
        feat, thr, left = t["feature"], t["threshold"], t["left_child"]
        leaf = t["leaf_value"]
        lines = [
            f"{type(self).__name__} of depth {self.depth}, "
            f"{feat.numel()} nodes"
        ]

        def rec(nid: int, indent: str):
            f = int(feat[nid])
            if f < 0:
                vals = [round(float(v), 6) for v in leaf[nid]]
                val = vals[0] if len(vals) == 1 else vals
                lines.append(f"{indent}Predict: {val}")
                return
            th = float(thr[nid])
            lines.append(f"{indent}If (feature {f} <= {th:.6g})")
            rec(int(left[nid]), indent + " ")
            lines.append(f"{indent}Else (feature {f} > {th:.6g})")
            rec(int(left[nid]) + 1, indent + " ")

        rec(0, " ")
        return "\n".join(lines)

    @property
    def featureImportances(self):
        """Gain-based normalized feature importances (MLlib
        ``featureImportances`` analog): per-feature sums of split gains."""
        fi = self._tree.get("feature_importance")
        if fi is None:
            fi = torch.zeros(self._num_features)
        return fi.cpu()

    def _set_tree(self, tree: dict, num_features: int):
        self._tree = tree
        self._num_features = num_features

    def _predict_values(self, x: torch.Tensor) -> torch.Tensor:
        t = self._tree
        return ops.tree_predict(
            x,
            t["feature"],
            t["threshold"],
            t["left_child"],
            t["leaf_value"],
            max_depth=64,
        )

    @property
    def depth(self) -> int:
        # derived: longest root-to-leaf path
        feature = self._tree["feature"].cpu()
        left = self._tree["left_child"].cpu()

        def d(i):
            if feature[i] < 0:
                return 0
            li = int(left[i])
            return 1 + max(d(li), d(li + 1))

        return d(0)

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path, extra={"numFeatures": self._num_features, **self._extra_meta()}
        )
        persistence.save_tensors(os.path.join(path, "data"), self._tree)

    def _extra_meta(self):
        return {}

    def _load_extra(self, path: str, meta: dict):
        self._tree = persistence.load_tensors(os.path.join(path, "data"))
        self._num_features = meta.get("numFeatures", -1)




def _fused_fit_stats(grad, w, comm):
    """Single fused stats pass for a tree fit prologue: ONE host sync and
    (when distributed) one sum + one max collective replace the five
    separate `.item()` syncs (total weight, zero-weight check, unit-weight
    check, per-channel quantization maxima, root totals)."""
    d = grad.shape[1]
    n = grad.shape[0]
    dev = grad.device
    sums = torch.cat([
        w.sum().reshape(1),
        (w == 0).sum().reshape(1).to(torch.float32),
        (w != 1).sum().reshape(1).to(torch.float32),
        torch.tensor([float(n)], device=dev),
        grad.sum(dim=0).reshape(-1),
    ])
    maxs = torch.cat([
        grad.abs().amax(dim=0).reshape(-1) if n else torch.zeros(d, device=dev),
        w.abs().amax().reshape(1) if n else torch.zeros(1, device=dev),
    ])
    if comm.is_distributed:
        comm.all_reduce_(sums)
        comm.all_reduce_(maxs, "max")
    sums_c = sums.cpu()
    maxs_c = maxs.cpu()
    return {
        "total_w": float(sums_c[0]),
        "has_zero": float(sums_c[1]) > 0,
        "all_one": float(sums_c[2]) == 0,
        "n_global": float(sums_c[3]),
        "gsum": sums_c[4:],
        "gmax": maxs_c[:d],
        "hmax": float(maxs_c[d]),
    }


def _grow_args(st):
    """(hess_is_count, gh_max, root_tot) for grow_tree from fused stats;
    root totals only valid when every row participates (no zero-weight
    mask)."""
    d = st["gmax"].numel()
    if st["all_one"]:
        gh_max = torch.cat([st["gmax"], torch.tensor([st["hmax"]])])
        root_tot = torch.cat([st["gsum"],
                              torch.tensor([st["total_w"]])])
    else:
        gh_max = torch.cat([st["gmax"], torch.tensor([st["hmax"], 1.0])])
        root_tot = torch.cat([st["gsum"],
                              torch.tensor([st["total_w"], st["n_global"]])])
    return st["all_one"], gh_max, (None if st["has_zero"] else root_tot)


class DecisionTreeRegressor(Regressor, _TreeParams):
    def _fit(self, dataset: TensorFrame) -> "DecisionTreeRegressionModel":
        x, y, w = self._extract_xyw(dataset)
        edges, bins = ensure_binned(dataset, x, self.getOrDefault("maxBins"))
        comm = get_comm()
        grad = (w * y).unsqueeze(1)
        st = _fused_fit_stats(grad, w, comm)
        gp = self._grow_params(st["total_w"])
        mask = w > 0 if st["has_zero"] else None
        hic, gh_max, root_tot = _grow_args(st)
        tp_out: list = []
        tree = grow_tree(bins, edges, grad, w, gp, comm, row_mask=mask,
                         train_pred_out=tp_out, hess_is_count=hic,
                         gh_max_in=gh_max, root_tot_in=root_tot)
        model = DecisionTreeRegressionModel()
        model._set_tree(tree, x.shape[1])
        # training-row predictions captured during growth (leaf scatter
        # instead of a post-hoc tree walk; GBM's margin update uses this)
        model._train_pred = tp_out[0].squeeze(1) if tp_out else None
        model._copy_cols_from(self)
        return model

    def _copy_cols_to(self, model):
        pass


class DecisionTreeRegressionModel(_TreeModelMixin, RegressionModel, _TreeParams):
    def _copy_cols_from(self, est):
        for p in ("featuresCol", "labelCol", "predictionCol"):
            self.set(p, est.getOrDefault(p))

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        return self._predict_values(features.float()).squeeze(1)


def fit_tree_forest(learner, edges, bins, labels, weights, comm=None,
                    subspaces=None, root_rows=None):
    """T independent ``DecisionTreeRegressor`` fits as ONE fused forest
    grow (tree_grower.grow_forest): the MI355X form of the reference's
    per-class / per-learner fit futures.  ``labels`` [N, T] (or [N, 1]
    broadcast); ``weights`` [N] shared or [N, T] per-tree, strictly
    positive on each tree's rows (``root_rows`` restricts a tree to a row
    subset — pass the rows with positive bag weight; without it callers
    must guarantee all-positive weights).  ``subspaces``: per-tree sorted
    feature-index tensors — trees are grown in the FULL feature space
    with banned features masked out of the split search (identical
    splits to a sliced fit) and their split ids remapped to the
    subspace-local space afterwards, so the returned models are
    indistinguishable from sliced fits.  Returns (models,
    train_pred [N, T]) — train_pred is zero outside a tree's rows.

    Caller contract: ``learner`` is a plain DecisionTreeRegressor with
    minWeightFractionPerNode == 0 (per-tree total weights would otherwise
    need per-tree thresholds)."""
    from ..parallel import get_comm as _gc
    from .tree_grower import grow_forest

    comm = comm or _gc()
    F = bins.shape[1]
    w2 = weights if weights.dim() == 2 else weights.unsqueeze(1)
    grads = (labels * w2).contiguous()
    T = max(grads.shape[1], w2.shape[1])
    if grads.shape[1] != T:
        grads = grads.expand(-1, T).contiguous()
    # one fused stats sync: slot-wise quantization maxima + unit check
    stats = torch.stack([
        grads.abs().max(),
        weights.abs().max(),
        (weights != 1).sum().to(torch.float32),
    ])
    if comm.is_distributed:
        comm.all_reduce_(stats, "max")  # max of the unit-check is fine: >0 anywhere means not-unit
    stats_c = stats.cpu()
    hic = float(stats_c[2]) == 0.0
    gh_max = torch.tensor(
        [float(stats_c[0]), float(stats_c[1])] + ([] if hic else [1.0])
    )
    fmasks = None
    if subspaces is not None and any(
        s.numel() != F for s in subspaces
    ):
        fmasks = torch.zeros(T, F, dtype=torch.float32, device=bins.device)
        for t, sub in enumerate(subspaces):
            fmasks[t, sub.to(bins.device)] = 1.0
    gp = learner._grow_params(1.0)  # minWeightFraction==0 ⇒ total unused
    tp_out: list = []
    trees = grow_forest(bins, edges, grads, weights, gp, comm,
                        hess_is_count=hic, train_pred_out=tp_out,
                        gh_max_in=gh_max, root_rows=root_rows,
                        feature_masks=fmasks)
    models = []
    for t, tree in enumerate(trees):
        nf = F
        if fmasks is not None:
            sub = subspaces[t]
            nf = int(sub.numel())
            feat = tree["feature"].long()
            sub_dev = sub.to(device=feat.device, dtype=torch.long)
            local = torch.searchsorted(sub_dev, feat.clamp_min(0))
            tree = dict(tree, feature=torch.where(
                feat >= 0, local, feat
            ).to(torch.int32))
        m = DecisionTreeRegressionModel()
        m._set_tree(tree, nf)
        m._train_pred = tp_out[0][:, t]
        m._copy_cols_from(learner)
        models.append(m)
    return models, tp_out[0]


def fit_class_tree_forest(learner, edges, bins, onehot, weights, k_classes,
                          comm=None, subspaces=None, root_rows=None):
    """T independent ``DecisionTreeClassifier`` fits (gini trees, D = K
    one-hot channels) as fused forest grows.  ``onehot`` [N, K];
    ``weights`` [N, T] per-tree bag weights (positive on each tree's
    ``root_rows``).  K + 2 channels per tree must fit the kernel's
    8-channel cap (K <= 6 with non-unit weights).  Members are chunked so
    the [N, chunk, K] gradient tensor stays modest."""
    from ..parallel import get_comm as _gc
    from .tree_grower import grow_forest

    comm = comm or _gc()
    F = bins.shape[1]
    N, K = onehot.shape
    T = weights.shape[1]
    gp = learner._grow_params(1.0)
    stats = torch.stack([weights.abs().max()])
    if comm.is_distributed:
        comm.all_reduce_(stats, "max")
    hmax = float(stats.cpu()[0])
    gh_max = torch.tensor([max(hmax, 1.0)] * K + [hmax, 1.0])

    models = []
    chunk = max(2, 16 // max(K // 3, 1))
    for s in range(0, T, chunk):
        sl = slice(s, min(s + chunk, T))
        w_sl = weights[:, sl]
        grads = (onehot.unsqueeze(1) * w_sl.unsqueeze(2)).contiguous()
        fmasks = None
        subs_sl = subspaces[sl] if subspaces is not None else None
        if subs_sl is not None and any(x.numel() != F for x in subs_sl):
            fmasks = torch.zeros(len(subs_sl), F, dtype=torch.float32,
                                 device=bins.device)
            for t, sub in enumerate(subs_sl):
                fmasks[t, sub.to(bins.device)] = 1.0
        trees = grow_forest(
            bins, edges, grads, w_sl.contiguous(), gp, comm,
            hess_is_count=False, gh_max_in=gh_max,
            root_rows=root_rows[sl] if root_rows is not None else None,
            feature_masks=fmasks,
        )
        for t, tree in enumerate(trees):
            nf = F
            if fmasks is not None:
                sub = subs_sl[t]
                nf = int(sub.numel())
                feat = tree["feature"].long()
                sub_dev = sub.to(device=feat.device, dtype=torch.long)
                local = torch.searchsorted(sub_dev, feat.clamp_min(0))
                tree = dict(tree, feature=torch.where(
                    feat >= 0, local, feat
                ).to(torch.int32))
            m = DecisionTreeClassificationModel()
            m._set_tree(tree, nf)
            m._num_classes = k_classes
            m._copy_cols_from(learner)
            models.append(m)
    return models


class DecisionTreeClassifier(ProbabilisticClassifier, _TreeParams):
    def _fit(self, dataset: TensorFrame) -> "DecisionTreeClassificationModel":
        x, y, w = self._extract_xyw(dataset)
        k = self._get_num_classes(dataset)
        comm = get_comm()
        k = int(comm.all_reduce_scalar(k, "max"))
        edges, bins = ensure_binned(dataset, x, self.getOrDefault("maxBins"))
        onehot = torch.zeros(x.shape[0], k, dtype=torch.float32, device=x.device)
        onehot.scatter_(1, y.long().unsqueeze(1), 1.0)
        grad = onehot * w.unsqueeze(1)
        st = _fused_fit_stats(grad, w, comm)
        gp = self._grow_params(st["total_w"])
        mask = w > 0 if st["has_zero"] else None
        hic, gh_max, root_tot = _grow_args(st)
        tree = grow_tree(bins, edges, grad, w, gp, comm, row_mask=mask,
                         hess_is_count=hic, gh_max_in=gh_max,
                         root_tot_in=root_tot)
        model = DecisionTreeClassificationModel()
        model._set_tree(tree, x.shape[1])
        model._num_classes = k
        model._copy_cols_from(self)
        return model


class DecisionTreeClassificationModel(
    _TreeModelMixin, ProbabilisticClassificationModel, _TreeParams
):
    def _copy_cols_from(self, est):
        for p in (
            "featuresCol",
            "labelCol",
            "predictionCol",
            "rawPredictionCol",
            "probabilityCol",
        ):
            self.set(p, est.getOrDefault(p))

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        # leaf values are class-probability vectors; raw = probabilities
        return self._predict_values(features.float())

    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        s = raw.sum(dim=1, keepdim=True).clamp_min(1e-12)
        raw /= s
        return raw

    def _extra_meta(self):
        return {"numClasses": self._num_classes}

    def _load_extra(self, path: str, meta: dict):
        super()._load_extra(path, meta)
        self._num_classes = meta.get("numClasses", -1)
