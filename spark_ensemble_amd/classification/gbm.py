"""GBMClassifier — gradient/newton boosting for classification.

Re-creates reference classification/GBMClassifier.scala:219-496:
  * losses {logloss, exponential, bernoulli} (default logloss); LogLoss has
    dim = K, so each round fits K base regressors on per-class
    pseudo-residuals (reference fits them in parallel Futures :377-411;
    here, for the built-in tree learner on full bags, all K trees grow
    LEVEL-SYNCHRONOUSLY in fused kernel launches — one histogram build and
    one RCCL all-reduce per level for the whole round,
    tree_grower.grow_forest; generic base learners fit sequentially),
  * init {prior, uniform}; binary + dim-1 prior -> constant log-odds model
    (reference :275-283),
  * newton pseudo-residuals with per-dim hessian floor 1e-2 and weights
    0.5 * h_j / sum(h_j) * w (:337-375),
  * dim-D line search by L-BFGS-B with bounds [0, inf) started at
    ones(dim) (:413-431); scalar-dim smooth losses (bernoulli,
    exponential) use the safeguarded-Newton search instead of Brent
    (same minimizer, fewer full-data evaluations — see
    boosting/line_search.py),
  * patience early stop identical to GBMRegressor (:451-479),
  * model predictRaw = init raw + sum_i sum_j w_ij f_ij(slice(x)); binary
    dim-1 maps s -> (-s, s) (:567-589); raw2probability delegates to the
    loss (:564-565).

Documented deviations: (1) the bag is resampled with seed + i per round
(the reference reuses the same seed each round); (2) for the exponential
loss the binary class-1 probability uses the margin s (p1 = 1/(1+e^{-2s})):
the reference feeds its (-s, s) raw vector's element 0 into the loss's
formula, which inverts the probability ordering relative to the margin —
we keep the mathematically consistent orientation.
"""

from __future__ import annotations

import os
from typing import List, Optional

import numpy as np
import torch

from .. import persistence
from ..boosting.gbm_params import GBMParams
from ..boosting.line_search import optimize_weight_1d, optimize_weight_nd
from ..boosting.losses import get_classification_loss
from ..ensemble.binning import BinnedDataset
from ..ensemble.utils import slice_features, subspace
from ..estimator import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
)
from ..frame import TensorFrame
from ..models.dummy import DummyClassificationModel, DummyClassifier
from ..params import ParamValidators
from ..parallel import get_comm
from ..utils.stats import dist_mean

SUPPORTED_LOSSES = ["logloss", "exponential", "bernoulli"]
SUPPORTED_INIT = ["uniform", "prior"]


class _GBMClassifierParams(GBMParams):
    def _declare_params(self):
        super()._declare_params()
        self.initStrategy = self._str_param(
            "initStrategy",
            "init prediction strategy: uniform|prior",
            ParamValidators.inArray(SUPPORTED_INIT),
        )
        self.loss = self._str_param(
            "loss",
            "loss function: " + "|".join(SUPPORTED_LOSSES),
            ParamValidators.inArray(SUPPORTED_LOSSES),
        )
        self.parallelism = self._int_param(
            "parallelism", "concurrent per-class fits", ParamValidators.gtEq(1)
        )
        self._setDefault(initStrategy="prior", loss="logloss", parallelism=1)

    def getLoss(self):
        return self.getOrDefault("loss")

    def setLoss(self, v):
        return self.set("loss", v)

    def setInitStrategy(self, v):
        return self.set("initStrategy", v)


class GBMClassifier(ProbabilisticClassifier, _GBMClassifierParams):
    def _default_base_learner(self):
        from ..models.tree import DecisionTreeRegressor

        return DecisionTreeRegressor()

    def _fit(self, dataset: TensorFrame) -> "GBMClassificationModel":
        from ..utils.instrumentation import Instrumentation

        instr = Instrumentation(self, dataset)
        instr.log_params(self, "loss", "numBaseLearners", "updates",
                         "learningRate", "optimizedWeights")
        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        seed = self.getOrDefault("seed")
        k_stages = self.getNumBaseLearners()
        lr_rate = self.getOrDefault("learningRate")
        use_newton = self.getOrDefault("updates") == "newton"
        optimized = self.getOrDefault("optimizedWeights")

        vcol = self.getOrNone("validationIndicatorCol")
        if vcol:
            vmask = dataset[vcol].bool()
            train = dataset.filter(~vmask)
            val = dataset.filter(vmask)
        else:
            train, val = dataset, None

        x, y, w = self._extract_xyw(train)
        n, num_features = x.shape
        device = x.device
        num_classes = int(comm.all_reduce_scalar(self._get_num_classes(train), "max"))
        binned = BinnedDataset(x, train)

        loss = get_classification_loss(self.getLoss(), num_classes)
        dim = loss.dim

        subspaces = [
            subspace(self.getSubspaceRatio(), num_features, seed + i)
            for i in range(k_stages)
        ]

        # -- init model (reference :275-288) -------------------------------
        init_strategy = self.getOrDefault("initStrategy")
        if init_strategy == "prior" and dim == 1 and num_classes == 2:
            prior = DummyClassifier().setStrategy("prior").fit(
                TensorFrame(features=x, label=y, weight=w)
            )
            p1 = float(prior._prob[1])
            logodds = torch.tensor([np.log(p1 / max(1.0 - p1, 1e-300))])
            init = DummyClassificationModel.from_raw(logodds, num_classes, num_features)
        else:
            init = DummyClassifier().setStrategy(init_strategy).fit(
                TensorFrame(features=x, label=y, weight=w)
            )

        # margins: first `dim` components of the init raw (the reference
        # carries numClasses-length arrays but only indices < dim are ever
        # read or updated)
        predictions = init.predictRaw(x)[:, :dim].contiguous()
        ylab = loss.encode_label(y)

        if val is not None:
            xv, yv, wv = self._extract_xyw(val)
            yvlab = loss.encode_label(yv)
            val_pred = init.predictRaw(xv)[:, :dim].contiguous()
            best_err = dist_mean(loss.loss(yvlab, val_pred), comm)
        else:
            best_err = 0.0

        models: List[List] = []
        weights: List[List[float]] = []
        i = 0
        v = 0

        # -- resume from a round-state checkpoint ----------
        from ..utils import checkpoint as ckpt

        ckpt_dir = self.getCheckpointDir()
        ck_fp = (
            ckpt.fingerprint(self, n, num_features, y, w) if ckpt_dir else None
        )
        ck_saved: set = set()
        resumed = ckpt.load_round_state(ckpt_dir, ck_fp)
        if resumed:
            r0, models, weights, extra = resumed
            r0 = min(r0, k_stages)
            models, weights = models[:r0], weights[:r0]
            for j, (ms, wts) in enumerate(zip(models, weights)):
                xs_j = slice_features(x, subspaces[j])
                wt_t = torch.tensor(wts, dtype=torch.float32, device=device)
                dirs = torch.stack([m.predict(xs_j) for m in ms], dim=1)
                predictions = predictions + dirs * wt_t.unsqueeze(0)
                if val is not None:
                    xvs_j = slice_features(xv, subspaces[j])
                    vdirs = torch.stack([m.predict(xvs_j) for m in ms], dim=1)
                    val_pred = val_pred + vdirs * wt_t.unsqueeze(0)
            best_err = extra.get("best_err", best_err)
            v = int(extra.get("v", 0))
            i = r0
            instr.log_named_value("resumed_from_round", r0)

        prev_sol = None  # warm start for the n-D stage-weight search
        instr.timers["setup_ms"] = instr.elapsed_ms()

        # fused-round eligibility.  FIT-LOCAL state (not attributes on
        # self): concurrent fits of one shared estimator instance (fold
        # tasks on the stream pool) must not see each other's cached
        # weight-positivity answer — a stale True would run the fused
        # grower on zero-weight rows.  The static part depends only on
        # params; the positivity check needs one sync, done lazily on
        # the first round.
        from ..models.tree import DecisionTreeRegressor

        fuse_state = {
            "static_ok": (
                type(learner) is DecisionTreeRegressor
                and learner.getOrDefault("minWeightFractionPerNode") == 0.0
            ),
            "w_ok": None,
        }

        while i < k_stages and v < self.getOrDefault("numRounds"):
            idx = subspaces[i]
            xs = binned.sliced_features(idx)

            bag_m = self.sample_weights(
                self.getReplacement(),
                self.getSubsampleRatio(),
                n,
                seed + i,
                device,
                None,
                comm.rank,
            )
            bag_w = bag_m * w

            newton = use_newton and loss.has_hessian
            g, h = loss.grad_hess_fused(ylab, predictions, want_hess=newton)
            if newton:
                h = h.clamp_min(1e-2)  # [N, dim]
                sum_h = (h * bag_m.unsqueeze(1)).sum(dim=0)
                comm.all_reduce_(sum_h)
                res_label = -g / h  # [N, dim]
                res_weight = 0.5 * h / sum_h.unsqueeze(0) * bag_w.unsqueeze(1)
            else:
                res_label = -g
                res_weight = bag_w.unsqueeze(1).expand(-1, dim)

            # K per-class base-regressor fits.  The reference runs these
            # as driver-side parallel futures (:377-411); here, when the
            # base learner is the built-in histogram tree, ALL K trees
            # grow level-synchronously in fused kernel launches (ONE
            # histogram build + ONE all-reduce per level for the whole
            # round — tree_grower.grow_forest).  Generic base learners
            # keep the sequential per-class path.
            directions = None
            if dim > 1 and self._can_fuse_round(learner, res_weight,
                                                fuse_state):
                from ..models.tree import fit_tree_forest

                f_edges, f_bins = binned.sliced_binned(
                    idx, learner.getOrDefault("maxBins")
                )
                imodels, directions = fit_tree_forest(
                    learner, f_edges, f_bins, res_label, res_weight, comm
                )
            else:
                from ..parallel.streams import parallel_fits

                def class_task(j):
                    def task():
                        fr = binned.fit_frame(
                            learner, res_label[:, j].contiguous(),
                            res_weight[:, j].contiguous(), idx, xs,
                        )
                        return self.fit_base_learner(
                            learner, fr, weight_col="weight"
                        )
                    return task

                imodels = parallel_fits(
                    [class_task(j) for j in range(dim)],
                    self.getOrDefault("parallelism"), warm_first=True,
                )

            if directions is None:
                def _dir(m):
                    tp = getattr(m, "_train_pred", None)
                    return tp if tp is not None and tp.shape[0] == xs.shape[0] \
                        else m.predict(xs)

                directions = torch.stack([_dir(m) for m in imodels], dim=1)

            if optimized:
                if dim == 1:
                    sol = [
                        optimize_weight_1d(
                            loss, ylab, predictions, directions, bag_w, comm,
                            self.getOrDefault("maxIter"), self.getOrDefault("tol"),
                        )
                    ]
                else:
                    sol = optimize_weight_nd(
                        loss, ylab, predictions, directions, bag_w, comm,
                        self.getOrDefault("maxIter"), self.getOrDefault("tol"),
                        x0=prev_sol,
                    ).tolist()
                    prev_sol = sol
            else:
                sol = [1.0] * dim
            iweights = [s * lr_rate for s in sol]

            models.append(imodels)
            weights.append(iweights)

            wt = torch.tensor(iweights, dtype=torch.float32, device=device)
            predictions = predictions + directions * wt.unsqueeze(0)

            from ..boosting import line_search as _ls

            if val is not None:
                xvs = slice_features(xv, idx)
                vdir = torch.stack([m.predict(xvs) for m in imodels], dim=1)
                val_pred = val_pred + vdir * wt.unsqueeze(0)
                err = dist_mean(loss.loss(yvlab, val_pred), comm)
                if best_err - err < self.getOrDefault("validationTol") * max(err, 0.01):
                    v += 1
                elif err < best_err:
                    best_err = err
                    v = 0
                instr.log_round(i, weight=float(iweights[0]), val_loss=err,
                                patience=v, ls_evals=_ls.LAST_EVALS)
            else:
                instr.log_round(i, weight=float(iweights[0]),
                                ls_evals=_ls.LAST_EVALS)
            interval = self.getCheckpointInterval()
            if ckpt_dir and interval > 0 and (i + 1) % interval == 0:
                ckpt.save_round_state(
                    ckpt_dir, i + 1, models, weights,
                    extra={"best_err": best_err, "v": v},
                    fingerprint=ck_fp, _saved_dirs=ck_saved,
                )
            i += 1

        keep = i - v
        instr.finish()
        # resume state is crash recovery only; a completed fit clears it
        ckpt.clear(ckpt_dir)
        model = GBMClassificationModel()
        model._init = init
        model._models = models[:keep]
        model._weights = weights[:keep]
        model._subspaces = [s for s in subspaces[:keep]]
        model._num_classes = num_classes
        model._dim = dim
        model._num_features = num_features
        model.set("loss", self.getLoss())
        for p in (
            "featuresCol", "labelCol", "predictionCol",
            "rawPredictionCol", "probabilityCol",
        ):
            model.set(p, self.getOrDefault(p))
        return model

    # ---- fold-vectorized fitting (OOF stacking fast path) ----------------
    def _can_fit_folds(self) -> bool:
        """Covers the default classification configuration: gradient
        updates, prior/uniform init, full bags, identity subspace,
        built-in tree learner, no validation/checkpointing."""
        from ..models.tree import DecisionTreeRegressor

        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        return (
            self.getOrDefault("updates") == "gradient"
            and self.getSubspaceRatio() >= 1.0
            and self.getSubsampleRatio() >= 1.0
            and not self.getReplacement()
            and self.getOrNone("validationIndicatorCol") is None
            and not self.getCheckpointDir()
            and type(learner) is DecisionTreeRegressor
            and learner.getOrDefault("minWeightFractionPerNode") == 0.0
        )

    def _fit_folds(self, dataset: TensorFrame, fold: torch.Tensor,
                   num_folds: int, include_full: bool = False):
        """num_folds leave-one-fold-out GBM classifier fits grown JOINTLY:
        every boosting round fuses ALL (fold, class) trees into one
        forest (per-tree root row sets = the fold's rows); the per-fold
        stage-weight searches run on fold-masked weights.  The init is
        the GLOBAL prior — matching the sequential path, whose
        DummyClassifier ignores instance weights exactly like the
        reference's SQL aggregates (DummyClassifier.scala:102-109)."""
        from ..boosting import line_search as _ls  # noqa: F401
        from ..models.tree import DecisionTreeRegressionModel
        from ..models.tree_grower import grow_forest

        comm = get_comm()
        learner = self.getOrNone("baseLearner") or self._default_base_learner()
        k_stages = self.getNumBaseLearners()
        lr_rate = self.getOrDefault("learningRate")
        optimized = self.getOrDefault("optimizedWeights")

        x, y, w = self._extract_xyw(dataset)
        n, num_features = x.shape
        device = x.device
        num_classes = int(
            comm.all_reduce_scalar(self._get_num_classes(dataset), "max")
        )
        binned = BinnedDataset(x, dataset)
        edges, bins = binned.get(int(learner.getOrDefault("maxBins")))

        loss = get_classification_loss(self.getLoss(), num_classes)
        dim = loss.dim

        # shared global init (see docstring)
        init_strategy = self.getOrDefault("initStrategy")
        if init_strategy == "prior" and dim == 1 and num_classes == 2:
            prior = DummyClassifier().setStrategy("prior").fit(
                TensorFrame(features=x, label=y, weight=w)
            )
            p1 = float(prior._prob[1])
            logodds = torch.tensor([np.log(p1 / max(1.0 - p1, 1e-300))])
            init = DummyClassificationModel.from_raw(
                logodds, num_classes, num_features
            )
        else:
            init = DummyClassifier().setStrategy(init_strategy).fit(
                TensorFrame(features=x, label=y, weight=w)
            )

        fold = fold.to(device)
        masks = [(fold != f) for f in range(num_folds)]
        if include_full:
            masks.append(torch.ones_like(fold, dtype=torch.bool))
        Fo = len(masks)
        mask_f = torch.stack([m.float() for m in masks], dim=1)  # [N, Fo]
        wmask = w.unsqueeze(1) * mask_f
        rows_per_fold = [m.nonzero(as_tuple=True)[0].to(torch.int32)
                         for m in masks]
        root_rows = [rows_per_fold[f] for f in range(Fo) for _ in range(dim)]

        ylab = loss.encode_label(y)  # [N, dim]
        init_raw = init.predictRaw(x)[:, :dim].contiguous()
        margins = init_raw.unsqueeze(1).expand(n, Fo, dim).contiguous()
        lab_e = ylab.unsqueeze(1).expand(n, Fo, dim).reshape(-1, dim) \
            .contiguous()

        gp = learner._grow_params(1.0)
        stats = torch.stack([wmask.max()])
        if comm.is_distributed:
            comm.all_reduce_(stats, "max")
        hmax = float(stats.cpu()[0])

        fold_trees: List[List[List]] = [[] for _ in range(Fo)]
        fold_weights: List[List[List[float]]] = [[] for _ in range(Fo)]
        prev_sol: List = [None] * Fo
        for i in range(k_stages):
            g = loss.grad_hess_fused(
                lab_e, margins.reshape(-1, dim), want_hess=False
            )[0].reshape(n, Fo, dim)
            grads_T = ((-g) * wmask.unsqueeze(2)).reshape(n, Fo * dim) \
                .contiguous()
            gmax = float(grads_T.abs().max())
            if comm.is_distributed:
                gmax = comm.all_reduce_scalar(gmax, "max")
            gh_max = torch.tensor([max(gmax, 1e-30), max(hmax, 1e-30), 1.0])
            hess_T = wmask.repeat_interleave(dim, dim=1).contiguous()
            tp: list = []
            trees = grow_forest(bins, edges, grads_T, hess_T, gp, comm,
                                hess_is_count=False, train_pred_out=tp,
                                gh_max_in=gh_max, root_rows=root_rows)
            d = tp[0].reshape(n, Fo, dim)
            wts_round = torch.ones(Fo, dim, device=device)
            for f in range(Fo):
                if optimized:
                    mf = margins[:, f].contiguous()
                    df = d[:, f].contiguous()
                    if dim == 1:
                        sol = [optimize_weight_1d(
                            loss, ylab, mf, df, wmask[:, f].contiguous(),
                            comm, self.getOrDefault("maxIter"),
                            self.getOrDefault("tol"),
                        )]
                    else:
                        sol = optimize_weight_nd(
                            loss, ylab, mf, df, wmask[:, f].contiguous(),
                            comm, self.getOrDefault("maxIter"),
                            self.getOrDefault("tol"), x0=prev_sol[f],
                        ).tolist()
                        prev_sol[f] = sol
                else:
                    sol = [1.0] * dim
                iweights = [s * lr_rate for s in sol]
                fold_weights[f].append(iweights)
                wts_round[f] = torch.tensor(iweights, device=device)
                ms = []
                for j in range(dim):
                    m = DecisionTreeRegressionModel()
                    m._set_tree(trees[f * dim + j], num_features)
                    m._copy_cols_from(learner)
                    ms.append(m)
                fold_trees[f].append(ms)
            margins = margins + d * wts_round.unsqueeze(0)

        ident = torch.arange(num_features)
        out = []
        for f in range(Fo):
            model = GBMClassificationModel()
            model._init = init
            model._models = fold_trees[f]
            model._weights = fold_weights[f]
            model._subspaces = [ident] * k_stages
            model._num_classes = num_classes
            model._dim = dim
            model._num_features = num_features
            model.set("loss", self.getLoss())
            for p in (
                "featuresCol", "labelCol", "predictionCol",
                "rawPredictionCol", "probabilityCol",
            ):
                model.set(p, self.getOrDefault(p))
            out.append(model)
        if include_full:
            return out[:-1], out[-1]
        return out

    def _can_fuse_round(self, learner, res_weight, fuse_state) -> bool:
        """Fused K-tree rounds need: built-in tree learner, no per-tree
        weight thresholds, and strictly positive weights on every row
        (the fused grower has no zero-weight row mask).  Sub-sampling
        introduces zeros, so only the full-bag configs fuse.
        ``fuse_state`` is fit-local (see _fit) so concurrent fits of a
        shared estimator cannot cross-contaminate the cached answer."""
        if not fuse_state.get("static_ok", False):
            return False
        if self.getSubsampleRatio() < 1.0 or self.getReplacement():
            return False
        if fuse_state["w_ok"] is None:
            fuse_state["w_ok"] = bool((res_weight > 0).all())
        return fuse_state["w_ok"]

    def _save_impl(self, path: str):
        persistence.save_metadata(self, path)
        self._save_learner(path)

    def _load_extra(self, path: str, meta: dict):
        self.setBaseLearner(self._load_learner(path))


class GBMClassificationModel(ProbabilisticClassificationModel, _GBMClassifierParams):
    @property
    def models(self):
        """Per-(stage, class) base models (reference .models)."""
        return [list(ms) for ms in self._models]

    @property
    def weights(self):
        """Per-(stage, class) weights (reference .weights)."""
        return [list(ws) for ws in self._weights]
    _init = None
    _models: List[List] = []
    _weights: List[List[float]] = []
    _subspaces: List[torch.Tensor] = []
    _dim: int = 1

    def _loss_obj(self):
        return get_classification_loss(self.getLoss(), self._num_classes)

    def _margins(self, x: torch.Tensor) -> torch.Tensor:
        from ..ensemble.utils import packed_forest_margin

        out = self._init.predictRaw(x)[:, : self._dim].contiguous()
        if self._models:
            # fast path: identity subspaces + tree stages -> one packed
            # forest_predict kernel per class dimension (arena cached
            # across transform calls)
            caches = self.__dict__.setdefault("_pack_caches", {})
            packed_ok = True
            for j in range(self._dim):
                pj = packed_forest_margin(
                    x, [ms[j] for ms in self._models],
                    [wts[j] for wts in self._weights],
                    self._subspaces, x.shape[1],
                    cache=caches.setdefault(j, {}),
                )
                if pj is None:
                    packed_ok = False
                    break
                out[:, j] += pj
            if packed_ok:
                return out
            out = self._init.predictRaw(x)[:, : self._dim].contiguous()
        for wts, sub, ms in zip(self._weights, self._subspaces, self._models):
            xs = slice_features(x, sub)
            for j, m in enumerate(ms):
                out[:, j] += wts[j] * m.predict(xs)
        return out

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        x = features.float()
        s = self._margins(x)
        if self._dim == 1 and self._num_classes == 2:
            return torch.cat([-s, s], dim=1)
        return s

    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        loss = self._loss_obj()
        name = loss.name
        if name == "logloss":
            return torch.softmax(raw, dim=1, out=raw)
        # binary margin losses: raw = (-s, s)
        s = raw[:, 1]
        if name == "exponential":
            # deviation from reference (see module docstring): use the margin
            p1 = 1.0 / (1.0 + torch.exp(-2.0 * s))
        else:  # bernoulli — reference formula on raw[0] = -s
            p1 = 1.0 / (1.0 + torch.exp(-s))
        raw[:, 0] = 1.0 - p1
        raw[:, 1] = p1
        return raw

    @property
    def numModels(self) -> int:
        return len(self._models)

    @property
    def featureImportances(self):
        from ..ensemble.utils import ensemble_feature_importances

        ms, ws, subs = [], [], []
        for mrow, wrow, sub in zip(self._models, self._weights, self._subspaces):
            for m, w in zip(mrow, wrow):
                ms.append(m); ws.append(w); subs.append(sub)
        return ensemble_feature_importances(ms, ws, subs, self._num_features)

    # -- persistence: two-level model-$i-$k nesting (reference :621-634) ---
    def _save_impl(self, path: str):
        persistence.save_metadata(
            self,
            path,
            extra={
                "numClasses": self._num_classes,
                "numModels": len(self._models),
                "dim": self._dim,
                "numFeatures": self._num_features,
            },
        )
        self._init.save(os.path.join(path, "init"), overwrite=True)
        for i, ms in enumerate(self._models):
            for k, m in enumerate(ms):
                m.save(os.path.join(path, f"model-{i}-{k}"), overwrite=True)
                persistence.save_json_rows(
                    os.path.join(path, f"data-{i}-{k}"),
                    [{
                        "weight": float(self._weights[i][k]),
                        "subspace": self._subspaces[i].tolist(),
                    }],
                )

    def _load_extra(self, path: str, meta: dict):
        self._num_classes = meta["numClasses"]
        self._dim = meta["dim"]
        self._num_features = meta.get("numFeatures", -1)
        self._init = persistence.load_instance(os.path.join(path, "init"))
        self._models = []
        self._weights = []
        self._subspaces = []
        i = 0
        while os.path.isdir(os.path.join(path, f"model-{i}-0")):
            ms, ws = [], []
            k = 0
            while os.path.isdir(os.path.join(path, f"model-{i}-{k}")):
                ms.append(
                    persistence.load_instance(os.path.join(path, f"model-{i}-{k}"))
                )
                row = persistence.load_json_rows(os.path.join(path, f"data-{i}-{k}"))[0]
                ws.append(row["weight"])
                if k == 0:
                    self._subspaces.append(
                        torch.tensor(row["subspace"], dtype=torch.long)
                    )
                k += 1
            self._models.append(ms)
            self._weights.append(ws)
            i += 1
