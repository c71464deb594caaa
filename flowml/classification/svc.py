"""LinearSVC (spark:.../ml/classification/LinearSVC.scala).

Binary linear SVM: hinge loss + L2 regularization, solved by LBFGS on the
subgradient in standardized-feature space (Spark drives breeze LBFGS/OWLQN
the same way). rawPrediction = [-margin, margin]; no probability column
(LinearSVC is a plain Classifier in Spark too).
"""

from __future__ import annotations

import torch

from flowml import dist
from flowml.classification.summary import (BinaryClassificationTrainingSummary,
                                           _HasTrainingSummary)
from flowml.classification.base import (
    Classifier,
    ClassificationModel,
    _features_tensor,
    _label_tensor,
)
from flowml.frame import FlowFrame
from flowml.param import HasMaxIter, HasTol, HasWeightCol, Param, TypeConverters


class _LinearSVCParams(HasMaxIter, HasTol, HasWeightCol):
    regParam: Param[float] = Param(
        "undefined", "regParam", "regularization parameter (>= 0).",
        TypeConverters.toFloat,
    )
    fitIntercept: Param[bool] = Param(
        "undefined", "fitIntercept", "whether to fit an intercept term.",
        TypeConverters.toBoolean,
    )
    standardization: Param[bool] = Param(
        "undefined", "standardization",
        "whether to standardize the training features before fitting.",
        TypeConverters.toBoolean,
    )
    threshold: Param[float] = Param(
        "undefined", "threshold",
        "threshold in binary classification applied to the rawPrediction.",
        TypeConverters.toFloat,
    )


class LinearSVC(_LinearSVCParams, Classifier):
    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(maxIter=100, regParam=0.0, tol=1e-6,
                         fitIntercept=True, standardization=True, threshold=0.0)
        self._set(**kwargs)

    def _fit(self, dataset: FlowFrame) -> "LinearSVCModel":
        from flowml.optim import lbfgs

        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.float32)
        dev = X.device
        N, F = X.shape
        if bool((y.max() > 1).item()) if N else False:
            raise ValueError("LinearSVC only supports binary labels {0, 1}")
        ysign = 2.0 * y - 1.0
        w = None
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float32)
        wcol = w if w is not None else torch.ones(N, device=dev)

        stats = torch.zeros(1 + 2 * F, dtype=torch.float64, device=dev)
        w64 = wcol.to(torch.float64)
        stats[0] = w64.sum()
        X64 = X.to(torch.float64)
        stats[1 : 1 + F] = (X64 * w64.unsqueeze(1)).sum(dim=0)
        stats[1 + F :] = (X64**2 * w64.unsqueeze(1)).sum(dim=0)
        dist.all_reduce_(stats)
        sumW = float(stats[0])
        mean = (stats[1 : 1 + F] / max(sumW, 1e-300)).to(torch.float32)
        var = (stats[1 + F :] / max(sumW, 1e-300)).to(torch.float32) - mean**2
        std = var.clamp_min(0).sqrt()
        inv_std = torch.where(std > 0, 1.0 / std, torch.zeros_like(std))
        Xs = X * inv_std

        fit_intercept = self.getOrDefault(self.fitIntercept)
        reg = self.getOrDefault(self.regParam)
        standardization = self.getOrDefault(self.standardization)
        l2_w = torch.ones(F, device=dev) if standardization else inv_std**2
        n_params = F + (1 if fit_intercept else 0)
        theta0 = torch.zeros(n_params, dtype=torch.float32, device=dev)

        def f_grad(theta: torch.Tensor):
            beta = theta[:F]
            b = theta[F] if fit_intercept else torch.zeros((), device=dev)
            margin = Xs @ beta + b
            viol = (1.0 - ysign * margin).clamp_min(0.0)
            active = (viol > 0).to(torch.float32) * wcol
            loss_s = (wcol * viol).sum()
            d = -ysign * active
            gW = d @ Xs
            gB = d.sum().reshape(1)
            buf = torch.cat([loss_s.reshape(1), gW, gB])
            dist.all_reduce_(buf)
            loss = float(buf[0]) / sumW
            gW = buf[1 : 1 + F] / sumW
            gB = buf[1 + F :] / sumW
            if reg > 0:
                loss += 0.5 * reg * float((l2_w * beta**2).sum())
                gW = gW + reg * l2_w * beta
            return loss, (torch.cat([gW, gB]) if fit_intercept else gW)

        theta, history = lbfgs(f_grad, theta0, max_iter=self.getMaxIter(),
                               tol=self.getTol())
        coef = (theta[:F] * inv_std).detach().cpu()
        intercept = float(theta[F]) if fit_intercept else 0.0
        model = LinearSVCModel(coefficients=coef, intercept=intercept,
                               numFeatures=F)
        self._copyValues(model)
        model._set_parent(self)
        # Spark LinearSVCTrainingSummary: binary curves scored on the raw
        # margin column (LinearSVC has no probability column)
        model._summary = BinaryClassificationTrainingSummary(
            model, dataset, self.getLabelCol(),
            self.getOrDefault(self.predictionCol),
            probabilityCol=self.getOrDefault(self.rawPredictionCol))
        return model


class LinearSVCModel(_LinearSVCParams, _HasTrainingSummary, ClassificationModel):
    def __init__(self, coefficients=None, intercept: float = 0.0, numFeatures: int = 0):
        super().__init__()
        if coefficients is None:
            self._coef = torch.zeros(0)
        elif isinstance(coefficients, torch.Tensor):
            self._coef = coefficients.to(torch.float32)
        else:
            import numpy as np

            self._coef = torch.from_numpy(np.asarray(coefficients, dtype=np.float32))
        self.intercept = float(intercept)
        self.numFeatures = numFeatures
        self.numClasses = 2

    @property
    def coefficients(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._coef.to(torch.float64).numpy())

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        c = self._coef.to(device=features.device, dtype=torch.float32)
        m = features.to(torch.float32) @ c + self.intercept
        return torch.stack([-m, m], dim=1)

    def raw2prediction(self, raw: torch.Tensor) -> torch.Tensor:
        thr = self.getOrDefault(self.threshold)
        return (raw[:, 1] > thr).to(torch.float64)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)
