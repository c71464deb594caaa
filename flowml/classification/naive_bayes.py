"""NaiveBayes (spark:.../ml/classification/NaiveBayes.scala).

modelType ∈ multinomial (default) | bernoulli | gaussian, with Laplace
``smoothing`` (default 1.0). Fit is a handful of per-class reductions over
the shard (all-reduced across ranks — SURVEY.md §2.5 C6 pattern); predictRaw
returns per-class log-likelihoods, probabilities via softmax of the raw
scores (Spark's ProbabilisticClassifier contract).
"""

from __future__ import annotations

import math

import torch

from flowml import dist
from flowml.classification.base import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    _features_tensor,
    _label_tensor,
)
from flowml.classification.summary import ClassificationTrainingSummary, _HasTrainingSummary
from flowml.frame import FlowFrame
from flowml.param import HasWeightCol, Param, TypeConverters


class _NaiveBayesParams(HasWeightCol):
    smoothing: Param[float] = Param(
        "undefined", "smoothing", "The smoothing parameter (>= 0, default 1.0).",
        TypeConverters.toFloat,
    )
    modelType: Param[str] = Param(
        "undefined", "modelType",
        "The model type: multinomial (default) | bernoulli | gaussian.",
        TypeConverters.toString,
    )


class NaiveBayes(_NaiveBayesParams, ProbabilisticClassifier):
    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(smoothing=1.0, modelType="multinomial")
        self._set(**kwargs)

    def _fit(self, dataset: FlowFrame) -> "NaiveBayesModel":
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float64)
        y = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.int64)
        dev = X.device
        N, F = X.shape
        mtype = self.getOrDefault(self.modelType)
        lam = self.getOrDefault(self.smoothing)
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64)
        else:
            w = torch.ones(N, dtype=torch.float64, device=dev)

        kmax = torch.tensor([float(y.max().item()) if N else 0.0], device=dev)
        if dist.is_initialized():
            torch.distributed.all_reduce(kmax, op=torch.distributed.ReduceOp.MAX)
        C = max(int(kmax.item()) + 1, 2)

        onehot = torch.zeros(N, C, dtype=torch.float64, device=dev)
        onehot[torch.arange(N, device=dev), y] = 1.0
        woh = onehot * w.unsqueeze(1)                       # N×C
        class_w = woh.sum(dim=0)                            # C
        if mtype in ("multinomial", "bernoulli"):
            if mtype == "bernoulli":
                bad = (X != 0) & (X != 1)
                if bool(bad.any()):
                    raise ValueError("Bernoulli NaiveBayes requires 0/1 features")
            feat_w = woh.T @ X                              # C×F (Σ w·x per class)
            buf = torch.cat([class_w, feat_w.reshape(-1)])
            dist.all_reduce_(buf)
            class_w = buf[:C]
            feat_w = buf[C:].view(C, F)
            pi = torch.log(class_w) - math.log(float(class_w.sum()))
            if mtype == "multinomial":
                denom = torch.log(feat_w.sum(dim=1) + lam * F).unsqueeze(1)
                theta = torch.log(feat_w + lam) - denom
                sigma = torch.zeros(0, dtype=torch.float64)
            else:
                denom = torch.log(class_w + 2.0 * lam).unsqueeze(1)
                theta = torch.log(feat_w + lam) - denom     # log P(x_j=1|c)
                sigma = torch.zeros(0, dtype=torch.float64)
        elif mtype == "gaussian":
            s1 = woh.T @ X                                  # C×F
            s2 = woh.T @ (X * X)
            buf = torch.cat([class_w, s1.reshape(-1), s2.reshape(-1)])
            dist.all_reduce_(buf)
            class_w = buf[:C]
            s1 = buf[C : C + C * F].view(C, F)
            s2 = buf[C + C * F :].view(C, F)
            pi = torch.log(class_w) - math.log(float(class_w.sum()))
            mu = s1 / class_w.unsqueeze(1)
            var = s2 / class_w.unsqueeze(1) - mu * mu
            # Spark adds epsilon = 1e-9 * max feature variance for stability
            eps = 1e-9 * float(var.max().clamp_min(0.0))
            theta = mu
            sigma = var.clamp_min(0.0) + max(eps, 1e-12)
        else:
            raise ValueError(f"Unsupported modelType {mtype!r}")

        model = NaiveBayesModel(
            pi=pi.cpu(), theta=theta.cpu(), sigma=sigma.cpu(),
            numClasses=C, numFeatures=F, modelType=mtype,
        )
        self._copyValues(model)
        model._set_parent(self)
        model._summary = ClassificationTrainingSummary(
            model, dataset, self.getLabelCol(), self.getPredictionCol()
        )
        return model


class NaiveBayesModel(_NaiveBayesParams, _HasTrainingSummary, ProbabilisticClassificationModel):
    def __init__(self, pi=None, theta=None, sigma=None, numClasses: int = 0,
                 numFeatures: int = 0, modelType: str = "multinomial"):
        super().__init__()
        self._pi = pi if pi is not None else torch.zeros(0)
        self._theta = theta if theta is not None else torch.zeros(0)
        self._sigma = sigma if sigma is not None else torch.zeros(0)
        self.numClasses = numClasses
        self.numFeatures = numFeatures
        self._mtype = modelType

    @property
    def pi(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._pi.numpy())

    @property
    def theta(self):
        from flowml.linalg import Matrices

        return Matrices.dense(self._theta.shape[0], self._theta.shape[1],
                              self._theta.numpy().T.flatten())

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        X = features.to(torch.float64)
        dev = X.device
        pi = self._pi.to(dev)
        th = self._theta.to(dev)
        if self._mtype == "multinomial":
            return X @ th.T + pi
        if self._mtype == "bernoulli":
            neg = torch.log1p(-torch.exp(th).clamp(max=1 - 1e-15))
            return X @ (th - neg).T + (pi + neg.sum(dim=1))
        sg = self._sigma.to(dev)
        # gaussian log-likelihood per class
        const = -0.5 * torch.log(2 * math.pi * sg).sum(dim=1)   # C
        x2 = X.unsqueeze(1)                                      # N×1×F
        ll = -0.5 * ((x2 - th.unsqueeze(0)) ** 2 / sg.unsqueeze(0)).sum(dim=2)
        return ll + const + pi

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        return torch.softmax(raw, dim=1)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)
