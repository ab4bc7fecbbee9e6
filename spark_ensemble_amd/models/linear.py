"""Linear base learners (LinearRegression / LogisticRegression).

Stand-ins for the MLlib linear models used as base learners by the
reference's users (BASELINE config 4: BoostingClassifier with
LogisticRegression base exercises this GEMM path).

MI355X design: the hot ops are plain GEMMs — X[N,F] @ W[F,K] for margins
and X^T @ G for gradients — which go through rocBLAS (library GEMMs are the
sanctioned path for un-fused matmuls); the per-row loss/gradient transform
between them is fused elementwise work.  Optimization is batched full-data
L-BFGS on the host driving GPU loss/grad evaluations (the MLlib analog:
``RDDLossFunction`` + Breeze LBFGS), with RCCL all-reduce of the
(loss, gradient) payload per evaluation — the same collective shape as the
reference's treeAggregate at GBMClassifier.scala:423-427.
"""

from __future__ import annotations

import numpy as np
import torch

from .. import persistence
from ..estimator import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    RegressionModel,
    Regressor,
)
from ..frame import TensorFrame
from ..params import Params, ParamValidators
from ..parallel import get_comm

import os


class _LinearParams(Params):
    def _declare_params(self):
        super()._declare_params()
        self.regParam = self._float_param(
            "regParam", "L2 regularization", ParamValidators.gtEq(0.0)
        )
        self.maxIter = self._int_param(
            "maxIter", "max optimizer iterations", ParamValidators.gtEq(0)
        )
        self.tol = self._float_param(
            "tol", "convergence tolerance", ParamValidators.gtEq(0.0)
        )
        self.fitIntercept = self._bool_param("fitIntercept", "fit an intercept term")
        self.standardization = self._bool_param(
            "standardization", "standardize features before fitting"
        )
        self._setDefault(
            regParam=0.0, maxIter=100, tol=1e-6, fitIntercept=True,
            standardization=False,
        )

    def setRegParam(self, v):
        return self.set("regParam", v)

    def setMaxIter(self, v):
        return self.set("maxIter", v)

    def setTol(self, v):
        return self.set("tol", v)

    def setFitIntercept(self, v):
        return self.set("fitIntercept", v)


class LinearRegression(Regressor, _LinearParams):
    """Weighted ridge regression by normal equations:
    (X^T W X + lambda I) beta = X^T W y, solved on-GPU; the Gram matrix is
    all-reduced so every rank solves the same [F+1]^2 system."""

    def _fit(self, dataset: TensorFrame) -> "LinearRegressionModel":
        x, y, w = self._extract_xyw(dataset)
        comm = get_comm()
        intercept = self.getOrDefault("fitIntercept")
        if intercept:
            ones = torch.ones(x.shape[0], 1, dtype=x.dtype, device=x.device)
            xa = torch.cat([x, ones], dim=1)
        else:
            xa = x
        f = xa.shape[1]
        xw = xa * w.unsqueeze(1)
        gram = xw.T @ xa  # [F, F]
        rhs = xw.T @ y  # [F]
        if comm.is_distributed:
            comm.all_reduce_(gram)
            comm.all_reduce_(rhs)
        lam = self.getOrDefault("regParam")
        reg = torch.eye(f, dtype=gram.dtype, device=gram.device) * (lam + 1e-8)
        if intercept:
            reg[f - 1, f - 1] = 1e-12  # do not regularize the intercept
        beta = torch.linalg.solve(gram.double() + reg.double(), rhs.double()).float()
        model = LinearRegressionModel()
        if intercept:
            model._coef = beta[:-1].contiguous()
            model._intercept = float(beta[-1])
        else:
            model._coef = beta.contiguous()
            model._intercept = 0.0
        model._num_features = x.shape[1]
        for p in ("featuresCol", "labelCol", "predictionCol"):
            model.set(p, self.getOrDefault(p))
        return model


class LinearRegressionModel(RegressionModel, _LinearParams):
    _coef: torch.Tensor
    _intercept: float = 0.0

    @property
    def coefficients(self):
        return self._coef

    @property
    def intercept(self):
        return self._intercept

    def predict(self, features: torch.Tensor) -> torch.Tensor:
        return features.float() @ self._coef.to(features.device) + self._intercept

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={"intercept": self._intercept, "numFeatures": self._num_features},
        )
        persistence.save_tensors(os.path.join(path, "data"), {"coef": self._coef})

    def _load_extra(self, path: str, meta: dict):
        self._intercept = float(meta["intercept"])
        self._num_features = meta.get("numFeatures", -1)
        self._coef = persistence.load_tensors(os.path.join(path, "data"))["coef"]


class LogisticRegression(ProbabilisticClassifier, _LinearParams):
    """Multinomial logistic regression via host L-BFGS over GPU-evaluated
    full-batch loss/gradient (GEMM + fused softmax-grad), with the
    (loss, grad) payload all-reduced across ranks per evaluation."""

    def _fit(self, dataset: TensorFrame) -> "LogisticRegressionModel":
        from scipy.optimize import minimize

        x, y, w = self._extract_xyw(dataset)
        comm = get_comm()
        k = int(comm.all_reduce_scalar(self._get_num_classes(dataset), "max"))
        n, f = x.shape
        intercept = self.getOrDefault("fitIntercept")
        lam = self.getOrDefault("regParam")
        yl = y.long()
        total_w = comm.all_reduce_scalar(float(w.sum()))

        fp = f + (1 if intercept else 0)

        from ..ops import dispatch

        y_int = yl.to(torch.int32)

        def eval_loss_grad(theta_np):
            theta = torch.from_numpy(theta_np.astype(np.float32)).to(x.device)
            wt = theta.view(fp, k)
            # single fused pass over X: margins + softmax loss + grad
            # outer-product (HIP kernel csrc/linear.hip on GPU)
            payload = dispatch.logreg_loss_grad(x, y_int, w, wt, intercept)
            if comm.is_distributed:
                comm.all_reduce_(payload)
            loss = float(payload[0]) / total_w
            grad_full = (payload[1:] / total_w).reshape(f + 1, k)
            grad = grad_full[:fp] if intercept else grad_full[:f]
            if lam > 0:
                reg_w = wt[:f]
                loss += 0.5 * lam * float((reg_w * reg_w).sum())
                grad[:f] += lam * reg_w
            return loss, grad.cpu().double().numpy().ravel()

        theta0 = np.zeros(fp * k)
        res = minimize(
            eval_loss_grad,
            theta0,
            jac=True,
            method="L-BFGS-B",
            options={
                "maxiter": self.getOrDefault("maxIter"),
                "ftol": self.getOrDefault("tol"),
                "gtol": self.getOrDefault("tol"),
            },
        )
        theta = torch.from_numpy(res.x.astype(np.float32)).view(fp, k)
        model = LogisticRegressionModel()
        model._coef = theta[:f].contiguous().to(x.device)
        model._intercept = (
            theta[f].contiguous().to(x.device)
            if intercept
            else torch.zeros(k, device=x.device)
        )
        model._num_classes = k
        model._num_features = f
        for p in (
            "featuresCol", "labelCol", "predictionCol",
            "rawPredictionCol", "probabilityCol",
        ):
            model.set(p, self.getOrDefault(p))
        return model


class LogisticRegressionModel(ProbabilisticClassificationModel, _LinearParams):
    _coef: torch.Tensor  # [F, K]
    _intercept: torch.Tensor  # [K]

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        return features.float() @ self._coef.to(features.device) + self._intercept.to(
            features.device
        )

    def raw2probabilityInPlace(self, raw: torch.Tensor) -> torch.Tensor:
        return torch.softmax(raw, dim=1, out=raw)

    def _save_impl(self, path: str):
        persistence.save_metadata(
            self, path,
            extra={"numClasses": self._num_classes, "numFeatures": self._num_features},
        )
        persistence.save_tensors(
            os.path.join(path, "data"),
            {"coef": self._coef, "intercept": self._intercept},
        )

    def _load_extra(self, path: str, meta: dict):
        self._num_classes = meta["numClasses"]
        self._num_features = meta.get("numFeatures", -1)
        d = persistence.load_tensors(os.path.join(path, "data"))
        self._coef = d["coef"]
        self._intercept = d["intercept"]
