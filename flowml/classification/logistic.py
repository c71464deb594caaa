"""LogisticRegression / LogisticRegressionModel.

Behavioral spec: ``spark:.../ml/classification/LogisticRegression.scala`` +
``optim/aggregator/LogisticAggregator.scala``: multinomial
softmax with ``family="auto"`` (binomial at 2 classes), training in
*standardized* feature space (``standardization=True`` default — coefficients
are transformed back after the fit), elastic-net regularization
(``regParam``, ``elasticNetParam``) with intercepts never penalized, LBFGS
(OWL-QN when L1 is active), ``maxIter=100``, ``tol=1e-6``.

MI355X mapping: the per-iteration treeAggregate of
LogisticAggregator becomes one fused ``all_reduce`` of the (K×(F+1)+1)-float
partial-gradient+loss buffer (C1, ~KB — latency-bound on xGMI); each rank
then takes the identical LBFGS step (replicated-state pattern, C4 elided).
The per-shard gradient itself is ``flowml.ops.logreg_grad`` (HIP kernel K1 on
GPU, torch reference on CPU).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from flowml import dist, ops
from flowml.classification.base import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    _features_tensor,
    _label_tensor,
)
from flowml.frame import FlowFrame
from flowml.classification.summary import (BinaryClassificationTrainingSummary,
                                           ClassificationTrainingSummary)
from flowml.linalg import DenseMatrix, DenseVector
from flowml.optim import lbfgs, owlqn
from flowml.param import (
    HasMaxIter,
    HasTol,
    HasWeightCol,
    Param,
    TypeConverters,
)


class _LogisticRegressionParams(HasMaxIter, HasTol, HasWeightCol):
    regParam: Param[float] = Param(
        "undefined", "regParam", "regularization parameter (>= 0).", TypeConverters.toFloat
    )
    elasticNetParam: Param[float] = Param(
        "undefined",
        "elasticNetParam",
        "the ElasticNet mixing parameter, in range [0, 1]. For alpha = 0, "
        "the penalty is an L2 penalty. For alpha = 1, it is an L1 penalty.",
        TypeConverters.toFloat,
    )
    fitIntercept: Param[bool] = Param(
        "undefined", "fitIntercept", "whether to fit an intercept term.", TypeConverters.toBoolean
    )
    standardization: Param[bool] = Param(
        "undefined",
        "standardization",
        "whether to standardize the training features before fitting the model.",
        TypeConverters.toBoolean,
    )
    family: Param[str] = Param(
        "undefined",
        "family",
        "The name of family which is a description of the label distribution to "
        "be used in the model. Supported options: auto, binomial, multinomial",
        TypeConverters.toString,
    )
    threshold: Param[float] = Param(
        "undefined",
        "threshold",
        "Threshold in binary classification prediction, in range [0, 1].",
        TypeConverters.toFloat,
    )
    aggregationDepth: Param[int] = Param(
        "undefined",
        "aggregationDepth",
        "suggested depth for treeAggregate (>= 2). (Kept for API parity; the "
        "RCCL all-reduce replaces treeAggregate — SURVEY.md §2.5 C1.)",
        TypeConverters.toInt,
    )

    def __init__(self):
        super().__init__()
        self._setDefault(
            maxIter=100,
            regParam=0.0,
            elasticNetParam=0.0,
            tol=1e-6,
            fitIntercept=True,
            standardization=True,
            family="auto",
            threshold=0.5,
            aggregationDepth=2,
        )

    def getRegParam(self) -> float:
        return self.getOrDefault(self.regParam)

    def getElasticNetParam(self) -> float:
        return self.getOrDefault(self.elasticNetParam)

    def getFitIntercept(self) -> bool:
        return self.getOrDefault(self.fitIntercept)

    def getStandardization(self) -> bool:
        return self.getOrDefault(self.standardization)

    def getFamily(self) -> str:
        return self.getOrDefault(self.family)

    def getThreshold(self) -> float:
        return self.getOrDefault(self.threshold)


class LogisticRegression(_LogisticRegressionParams, ProbabilisticClassifier):
    def __init__(
        self,
        *,
        featuresCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        predictionCol: Optional[str] = None,
        rawPredictionCol: Optional[str] = None,
        probabilityCol: Optional[str] = None,
        maxIter: Optional[int] = None,
        regParam: Optional[float] = None,
        elasticNetParam: Optional[float] = None,
        tol: Optional[float] = None,
        fitIntercept: Optional[bool] = None,
        standardization: Optional[bool] = None,
        family: Optional[str] = None,
        threshold: Optional[float] = None,
        weightCol: Optional[str] = None,
        aggregationDepth: Optional[int] = None,
    ):
        super().__init__()
        self._set(
            featuresCol=featuresCol,
            labelCol=labelCol,
            predictionCol=predictionCol,
            rawPredictionCol=rawPredictionCol,
            probabilityCol=probabilityCol,
            maxIter=maxIter,
            regParam=regParam,
            elasticNetParam=elasticNetParam,
            tol=tol,
            fitIntercept=fitIntercept,
            standardization=standardization,
            family=family,
            threshold=threshold,
            weightCol=weightCol,
            aggregationDepth=aggregationDepth,
        )

    # setters (pyspark style)
    def setRegParam(self, v):
        return self.set(self.regParam, v)

    def setElasticNetParam(self, v):
        return self.set(self.elasticNetParam, v)

    def setFamily(self, v):
        return self.set(self.family, v)

    def setFitIntercept(self, v):
        return self.set(self.fitIntercept, v)

    def setStandardization(self, v):
        return self.set(self.standardization, v)

    def setThreshold(self, v):
        return self.set(self.threshold, v)

    def _fit(self, dataset: FlowFrame) -> "LogisticRegressionModel":
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.int64)
        dev = X.device
        N, F = X.shape
        w = None
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float32)

        # global statistics across ranks (C6): numClasses, sumW, feature moments
        k_local = float(y.max().item()) if N else -1.0
        stats = torch.zeros(3 + 2 * F, dtype=torch.float64, device=dev)
        wcol = w if w is not None else torch.ones(N, device=dev)
        stats[0] = float(wcol.sum())
        stats[2] = N
        xw = X.to(torch.float64) * wcol.to(torch.float64).unsqueeze(1)
        stats[3 : 3 + F] = xw.sum(dim=0)
        stats[3 + F :] = (X.to(torch.float64) ** 2 * wcol.to(torch.float64).unsqueeze(1)).sum(dim=0)
        dist.all_reduce_(stats)
        kmax = torch.tensor([k_local], dtype=torch.float64, device=dev)
        if dist.is_initialized():
            torch.distributed.all_reduce(kmax, op=torch.distributed.ReduceOp.MAX)
        num_classes = max(int(kmax.item()) + 1, 2)
        sumW = float(stats[0].item())
        totalN = float(stats[2].item())
        mean = (stats[3 : 3 + F] / max(sumW, 1e-300)).to(torch.float32)
        var = (stats[3 + F :] / max(sumW, 1e-300)).to(torch.float32) - mean**2
        # Spark Summarizer std is the weighted sample std; with unit weights
        # this matches n/(n-1) correction:
        if totalN > 1:
            var = var * (totalN / (totalN - 1.0))
        std = var.clamp_min(0).sqrt()

        family = self.getFamily()
        if family == "auto":
            family = "binomial" if num_classes == 2 else "multinomial"
        is_multinomial = family == "multinomial"

        inv_std = torch.where(std > 0, 1.0 / std, torch.zeros_like(std))
        Xs = X * inv_std  # train in standardized space (no centering)

        fit_intercept = self.getFitIntercept()
        reg = self.getRegParam()
        alpha = self.getElasticNetParam()
        standardization = self.getStandardization()

        # per-coordinate L2/L1 multipliers in *scaled* space: scaled coef
        # beta_j corresponds to original w_j = beta_j * inv_std_j; if
        # standardization=False, Spark penalizes the ORIGINAL coefficients.
        l2_w = torch.ones(F, device=dev)
        l1_w = torch.ones(F, device=dev)
        if not standardization:
            l2_w = inv_std**2
            l1_w = inv_std

        K = num_classes if is_multinomial else 1

        def unpack(theta: torch.Tensor):
            Wc = theta[: K * F].view(K, F)
            bc = theta[K * F :] if fit_intercept else torch.zeros(K, device=dev)
            return Wc, bc

        n_params = K * F + (K if fit_intercept else 0)
        theta0 = torch.zeros(n_params, dtype=torch.float32, device=dev)
        if fit_intercept:
            # Spark initializes intercepts to log of label histogram ratios
            hist = torch.bincount(y, weights=wcol.to(torch.float64), minlength=num_classes).to(dev)
            dist.all_reduce_(hist)
            p = (hist + 1e-15) / (hist.sum() + num_classes * 1e-15)
            if is_multinomial:
                theta0[K * F :] = torch.log(p).to(torch.float32)
                theta0[K * F :] -= theta0[K * F :].mean()
            else:
                theta0[K * F] = float(torch.log(p[1] / p[0]))

        l2_coef = reg * (1.0 - alpha)
        l1_coef = reg * alpha

        def f_grad(theta: torch.Tensor):
            Wc, bc = unpack(theta)
            if is_multinomial:
                loss_s, gW, gB = ops.logreg_grad(Xs, y, w, Wc, bc)
            else:
                z = Xs @ Wc[0] + bc[0]
                yf = y.to(torch.float32)
                # numerically stable logistic loss
                nll = torch.nn.functional.softplus(z) - yf * z
                sig = torch.sigmoid(z)
                d = sig - yf
                if w is not None:
                    loss_s = (nll * w).sum()
                    d = d * w
                else:
                    loss_s = nll.sum()
                gW = (d @ Xs).unsqueeze(0)
                gB = d.sum().reshape(1)
            buf = torch.cat([loss_s.reshape(1), gW.reshape(-1), gB.reshape(-1)])
            dist.all_reduce_(buf)
            loss = float(buf[0]) / sumW
            gW = buf[1 : 1 + K * F].view(K, F) / sumW
            gB = buf[1 + K * F :] / sumW
            if l2_coef > 0:
                loss += 0.5 * l2_coef * float((l2_w * (Wc**2)).sum())
                gW = gW + l2_coef * l2_w * Wc
            grad = torch.cat([gW.reshape(-1), gB] if fit_intercept else [gW.reshape(-1)])
            return loss, grad

        max_iter = self.getMaxIter()
        tol = self.getTol()
        if l1_coef > 0:
            l1_vec = torch.zeros(n_params, device=dev)
            l1_vec[: K * F] = (l1_coef * l1_w).repeat(K)
            theta, history = owlqn(f_grad, theta0, l1_vec, max_iter=max_iter, tol=tol)
        else:
            theta, history = lbfgs(f_grad, theta0, max_iter=max_iter, tol=tol)

        Wc, bc = unpack(theta)
        W_orig = Wc * inv_std  # back to original feature space
        b_orig = bc.clone() if fit_intercept else torch.zeros(K, device=dev)

        model = LogisticRegressionModel(
            coefficients=W_orig.detach().cpu(),
            intercepts=b_orig.detach().cpu(),
            numClasses=num_classes,
            numFeatures=F,
            isMultinomial=is_multinomial,
        )
        self._copyValues(model)
        model._set_parent(self)
        model._summary = LogisticRegressionTrainingSummary(
            objectiveHistory=[float(h) for h in history],
            totalIterations=max(len(history) - 1, 0),
            model=model,
            dataset=dataset,
            labelCol=self.getLabelCol(),
            predictionCol=self.getPredictionCol(),
        )
        from flowml.utils.logging import get_logger

        get_logger().log(
            "lr_fit", iterations=model._summary.totalIterations,
            final_objective=float(history[-1]) if history else None,
            num_classes=num_classes, num_features=F,
        )
        return model


class LogisticRegressionModel(ProbabilisticClassificationModel, _LogisticRegressionParams):
    def __init__(
        self,
        coefficients: Optional[torch.Tensor] = None,  # K×F (K=1 binomial)
        intercepts: Optional[torch.Tensor] = None,  # K
        numClasses: int = 2,
        numFeatures: int = 0,
        isMultinomial: bool = False,
    ):
        super().__init__()
        self._W = coefficients if coefficients is not None else torch.zeros(1, 0)
        self._b = intercepts if intercepts is not None else torch.zeros(1)
        self.numClasses = numClasses
        self.numFeatures = numFeatures
        self.isMultinomial = isMultinomial
        self._summary: Optional[LogisticRegressionTrainingSummary] = None

    # -- pyspark-parity accessors -------------------------------------------

    @property
    def coefficientMatrix(self) -> DenseMatrix:
        W = self._W.to(torch.float64).cpu().numpy()
        return DenseMatrix(W.shape[0], W.shape[1], W.T.ravel())  # column-major

    @property
    def interceptVector(self) -> DenseVector:
        return DenseVector(self._b.to(torch.float64).cpu().numpy())

    @property
    def coefficients(self) -> DenseVector:
        if self.isMultinomial:
            raise RuntimeError("Multinomial models use coefficientMatrix")
        return DenseVector(self._W[0].to(torch.float64).cpu().numpy())

    @property
    def intercept(self) -> float:
        if self.isMultinomial:
            raise RuntimeError("Multinomial models use interceptVector")
        return float(self._b[0])

    @property
    def summary(self) -> "LogisticRegressionTrainingSummary":
        if self._summary is None:
            raise RuntimeError("No training summary available for this model")
        return self._summary

    def hasSummary(self) -> bool:
        return self._summary is not None

    # -- prediction ----------------------------------------------------------

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        W = self._W.to(device=features.device, dtype=torch.float32)
        b = self._b.to(device=features.device, dtype=torch.float32)
        m = features.to(torch.float32) @ W.T + b
        if self.isMultinomial:
            return m
        return torch.cat([-m, m], dim=1)  # Spark binary raw = [-margin, margin]

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        if self.isMultinomial:
            return torch.softmax(raw, dim=1)
        p1 = torch.sigmoid(raw[:, 1])
        return torch.stack([1 - p1, p1], dim=1)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        out = super()._transform(dataset)
        # binary threshold semantics (Spark: predict 1 iff p1 > threshold)
        thr = self.getThreshold()
        pred_col = self.getOrDefault(self.predictionCol)
        prob_col = self.getOrDefault(self.probabilityCol)
        if not self.isMultinomial and pred_col and thr != 0.5 and prob_col:
            p = out.col(prob_col)
            out = out.withColumn(pred_col, (p[:, 1] > thr).to(torch.float64))
        return out

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


class LogisticRegressionTrainingSummary(BinaryClassificationTrainingSummary):
    """Mirror of Spark's training summary: ``objectiveHistory`` /
    ``totalIterations`` plus the shared training-set metrics — and, for
    BINARY models, the threshold curves (roc/pr/…ByThreshold) of
    BinaryLogisticRegressionTrainingSummary. Curve accessors raise on
    multiclass models, like Spark's summary split."""

    def __init__(self, objectiveHistory: List[float], totalIterations: int,
                 model=None, dataset=None, labelCol: str = "label",
                 predictionCol: str = "prediction",
                 probabilityCol: str = "probability"):
        super().__init__(model, dataset, labelCol, predictionCol,
                         probabilityCol)
        self.objectiveHistory = objectiveHistory
        self.totalIterations = totalIterations

    def _curves(self):
        if getattr(self._model, "numClasses", 2) != 2:
            raise RuntimeError(
                "binary-only metric: this LogisticRegressionModel is "
                "multiclass (Spark exposes these on "
                "BinaryLogisticRegressionTrainingSummary only)")
        return super()._curves()
