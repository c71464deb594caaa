"""Factorization machines (spark:.../ml/classification/FMClassifier.scala,
regression/FMRegressor.scala).

Model: ŷ(x) = w₀ + Σᵢ wᵢxᵢ + ½ Σ_f [(Σᵢ v_{if} xᵢ)² − Σᵢ v_{if}² xᵢ²]
(the O(N·F·k) pairwise-interaction identity). Spark trains with mini-batch
gradient descent / adamW on the shard; here Adam (with coupled L2 — the
regParam enters through the gradient) runs FULL-batch —
the gradient is a handful of GEMMs over the resident shard (rocBLAS on GPU),
all-reduced across ranks like every other fit (SURVEY.md §2.5 C1 pattern).
Binary classification uses the logistic loss on ±1 labels, regression the
squared loss — Spark's choices.
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from flowml import dist
from flowml.base import Estimator
from flowml.classification.base import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    _features_tensor,
    _label_tensor,
)
from flowml.param import HasMaxIter, HasSeed, HasTol, Param, TypeConverters


class _FMParams(HasMaxIter, HasSeed, HasTol):
    factorSize: Param[int] = Param(
        "undefined", "factorSize",
        "Dimensionality of the factor vectors (>= 0; 0 = linear model).",
        TypeConverters.toInt,
    )
    fitLinear: Param[bool] = Param(
        "undefined", "fitLinear", "whether to fit the linear (1-way) term",
        TypeConverters.toBoolean,
    )
    fitIntercept: Param[bool] = Param(
        "undefined", "fitIntercept", "whether to fit an intercept term",
        TypeConverters.toBoolean,
    )
    regParam: Param[float] = Param(
        "undefined", "regParam", "L2 regularization parameter (>= 0)",
        TypeConverters.toFloat,
    )
    stepSize: Param[float] = Param(
        "undefined", "stepSize", "initial step size for the Adam solver (> 0)",
        TypeConverters.toFloat,
    )
    initStd: Param[float] = Param(
        "undefined", "initStd", "standard deviation of initial factors",
        TypeConverters.toFloat,
    )


def _fm_raw(X: torch.Tensor, w0: torch.Tensor, w: torch.Tensor,
            V: torch.Tensor) -> torch.Tensor:
    """ŷ per row via the O(NFk) identity; V is F×k."""
    out = w0 + X @ w
    if V.numel():
        xv = X @ V                       # N×k
        x2v2 = (X * X) @ (V * V)         # N×k
        out = out + 0.5 * (xv * xv - x2v2).sum(dim=1)
    return out


def _fit_fm(est, X: torch.Tensor, target: torch.Tensor, loss_kind: str):
    """Shared full-batch Adam loop. target: ±1 (logistic) or y (squared)."""
    dev = X.device
    N, F = X.shape
    k = est.getOrDefault(est.factorSize)
    fit_lin = est.getOrDefault(est.fitLinear)
    fit_b = est.getOrDefault(est.fitIntercept)
    reg = est.getOrDefault(est.regParam)
    lr = est.getOrDefault(est.stepSize)
    seed = est.getOrDefault(est.seed)
    gen = torch.Generator().manual_seed(seed & 0x7FFFFFFF)
    w0 = torch.zeros(1, device=dev)
    w = torch.zeros(F, device=dev)
    V = (torch.randn(F, k, generator=gen) * est.getOrDefault(est.initStd)).to(dev) \
        if k > 0 else torch.zeros(F, 0, device=dev)
    n_total = torch.tensor([float(N)], dtype=torch.float64, device=dev)
    dist.all_reduce_(n_total)
    n_all = float(n_total)
    # Adam state
    params = [w0, w, V]
    m = [torch.zeros_like(p) for p in params]
    v = [torch.zeros_like(p) for p in params]
    b1, b2, eps = 0.9, 0.999, 1e-8
    tol = est.getOrDefault(est.tol)
    prev = float("inf")
    for t in range(1, est.getMaxIter() + 1):
        yhat = _fm_raw(X, w0, w, V)
        if loss_kind == "logistic":
            z = target * yhat  # target ±1
            loss_s = torch.nn.functional.softplus(-z).sum()
            dl = -target * torch.sigmoid(-z)          # ∂loss/∂ŷ per row
        else:
            r = yhat - target
            loss_s = 0.5 * (r * r).sum()
            dl = r
        g0 = dl.sum().reshape(1)
        gw = dl @ X
        if k > 0:
            xv = X @ V
            gV = X.T @ (dl.unsqueeze(1) * xv) - (X * X).T @ dl.unsqueeze(1) * V
        else:
            gV = torch.zeros_like(V)
        buf = torch.cat([loss_s.reshape(1), g0, gw, gV.reshape(-1)]).to(torch.float64)
        dist.all_reduce_(buf)
        loss = float(buf[0]) / n_all
        g0 = (buf[1:2] / n_all).to(torch.float32)
        gw = (buf[2 : 2 + F] / n_all).to(torch.float32)
        gV = (buf[2 + F :].view(F, k) / n_all).to(torch.float32) if k > 0 else gV
        grads = [g0 if fit_b else torch.zeros_like(w0),
                 gw if fit_lin else torch.zeros_like(w),
                 gV]
        for i, (p, g) in enumerate(zip(params, grads)):
            if reg > 0 and i > 0:
                g = g + reg * p
            m[i].mul_(b1).add_(g, alpha=1 - b1)
            v[i].mul_(b2).addcmul_(g, g, value=1 - b2)
            mh = m[i] / (1 - b1**t)
            vh = v[i] / (1 - b2**t)
            p.add_(-lr * mh / (vh.sqrt() + eps))
        if abs(prev - loss) / max(abs(prev), abs(loss), 1e-12) < tol:
            break
        prev = loss
    return w0.detach().cpu(), w.detach().cpu(), V.detach().cpu()


class FMClassifier(_FMParams, ProbabilisticClassifier):
    """Binary FM classifier (logistic loss on ±1 labels, Adam)."""

    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(factorSize=8, fitLinear=True, fitIntercept=True,
                         regParam=0.0, stepSize=1.0, initStd=0.01,
                         maxIter=100, tol=1e-6, seed=42)
        self._set(**kwargs)

    def _fit(self, dataset) -> "FMClassificationModel":
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol()).to(torch.float32)
        if bool((y.max() > 1).item()) if len(y) else False:
            raise ValueError("FMClassifier only supports binary labels {0, 1}")
        w0, w, V = _fit_fm(self, X, 2.0 * y - 1.0, "logistic")
        model = FMClassificationModel(intercept=float(w0), linear=w, factors=V,
                                      numFeatures=X.shape[1])
        self._copyValues(model)
        model._set_parent(self)
        return model


class FMClassificationModel(_FMParams, ProbabilisticClassificationModel):
    def __init__(self, intercept: float = 0.0, linear: Optional[torch.Tensor] = None,
                 factors: Optional[torch.Tensor] = None, numFeatures: int = 0):
        super().__init__()
        self.intercept = float(intercept)
        self._linear = linear if linear is not None else torch.zeros(0)
        self._factors = factors if factors is not None else torch.zeros(0, 0)
        self.numFeatures = numFeatures
        self.numClasses = 2

    @property
    def linear(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._linear.to(torch.float64).numpy())

    @property
    def factors(self):
        from flowml.linalg import Matrices

        f = self._factors.to(torch.float64)
        return Matrices.dense(f.shape[0], f.shape[1], f.numpy().T.flatten())

    def _margin(self, features: torch.Tensor) -> torch.Tensor:
        X = features.to(torch.float32)
        return _fm_raw(X, torch.tensor(self.intercept, device=X.device),
                       self._linear.to(X.device), self._factors.to(X.device))

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        mgn = self._margin(features)
        return torch.stack([-mgn, mgn], dim=1)

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        p1 = torch.sigmoid(raw[:, 1])
        return torch.stack([1 - p1, p1], dim=1)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


from flowml.classification.base import _PredictorParams  # noqa: E402
from flowml.regression import RegressionModel, _RegressorIOMixin  # noqa: E402


class FMRegressor(_FMParams, _PredictorParams, Estimator, _RegressorIOMixin):
    """FM regressor (squared loss, Adam)."""

    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(factorSize=8, fitLinear=True, fitIntercept=True,
                         regParam=0.0, stepSize=1.0, initStd=0.01,
                         maxIter=100, tol=1e-6, seed=42)
        self._set(**kwargs)

    def _fit(self, dataset) -> "FMRegressionModel":
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol()).to(torch.float32)
        w0, w, V = _fit_fm(self, X, y, "squared")
        model = FMRegressionModel(intercept=float(w0), linear=w, factors=V,
                                  numFeatures=X.shape[1])
        self._copyValues(model)
        model._set_parent(self)
        return model


class FMRegressionModel(_FMParams, RegressionModel, _RegressorIOMixin):
    def __init__(self, intercept: float = 0.0, linear: Optional[torch.Tensor] = None,
                 factors: Optional[torch.Tensor] = None, numFeatures: int = 0):
        super().__init__()
        self.intercept = float(intercept)
        self._linear = linear if linear is not None else torch.zeros(0)
        self._factors = factors if factors is not None else torch.zeros(0, 0)
        self.numFeatures = numFeatures

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        X = features.to(torch.float32)
        return _fm_raw(X, torch.tensor(self.intercept, device=X.device),
                       self._linear.to(X.device), self._factors.to(X.device))
