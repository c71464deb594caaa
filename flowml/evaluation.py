"""MulticlassClassificationEvaluator.

Behavioral spec: ``spark:.../ml/evaluation/MulticlassClassificationEvaluator.scala``
(SURVEY.md §2.2): default ``metricName="f1"`` (the *weighted* F1), computed
from the (prediction, label) confusion matrix. ``macroF1`` is a flowml
extension — it is the task's headline metric (BASELINE.json:2).

Confusion counts are accumulated with a single ``bincount`` on-device, so the
evaluator runs on GPU tensors without a host round-trip.
"""

from __future__ import annotations

from typing import Optional

import torch

from flowml.frame import FlowFrame
from flowml.param import (
    HasLabelCol,
    HasPredictionCol,
    HasRawPredictionCol,
    HasWeightCol,
    Param,
    Params,
    TypeConverters,
)

_SUPPORTED = {
    "f1",
    "accuracy",
    "weightedPrecision",
    "weightedRecall",
    "weightedFMeasure",
    "truePositiveRateByLabel",
    "falsePositiveRateByLabel",
    "precisionByLabel",
    "recallByLabel",
    "fMeasureByLabel",
    "weightedTruePositiveRate",
    "weightedFalsePositiveRate",
    "hammingLoss",
    "logLoss",
    "macroF1",  # flowml extension (headline metric)
}


class MulticlassClassificationEvaluator(HasLabelCol, HasPredictionCol, HasWeightCol, Params):
    """Confusion-matrix multiclass metrics (weighted f1 default).

    >>> import torch
    >>> from flowml.frame import FlowFrame
    >>> df = FlowFrame({"prediction": torch.tensor([0.0, 1.0, 1.0, 0.0]),
    ...                 "label":      torch.tensor([0.0, 1.0, 0.0, 0.0])})
    >>> MulticlassClassificationEvaluator(metricName="accuracy").evaluate(df)
    0.75
    """

    metricName: Param[str] = Param(
        "undefined",
        "metricName",
        f"metric name in evaluation {sorted(_SUPPORTED)}",
        TypeConverters.toString,
    )
    metricLabel: Param[float] = Param(
        "undefined",
        "metricLabel",
        "The class whose metric will be computed in byLabel metrics.",
        TypeConverters.toFloat,
    )
    beta: Param[float] = Param(
        "undefined", "beta", "beta value in weightedFMeasure|fMeasureByLabel", TypeConverters.toFloat
    )
    probabilityCol: Param[str] = Param(
        "undefined", "probabilityCol", "probability column (for logLoss)", TypeConverters.toString
    )
    eps: Param[float] = Param(
        "undefined", "eps", "probability clamp for logLoss", TypeConverters.toFloat
    )

    def __init__(
        self,
        *,
        predictionCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        metricName: Optional[str] = None,
        metricLabel: Optional[float] = None,
        beta: Optional[float] = None,
        weightCol: Optional[str] = None,
        probabilityCol: Optional[str] = None,
        eps: Optional[float] = None,
    ):
        super().__init__()
        self._setDefault(
            predictionCol="prediction",
            labelCol="label",
            metricName="f1",
            metricLabel=0.0,
            beta=1.0,
            probabilityCol="probability",
            eps=1e-15,
        )
        self._set(
            predictionCol=predictionCol,
            labelCol=labelCol,
            metricName=metricName,
            metricLabel=metricLabel,
            beta=beta,
            weightCol=weightCol,
            probabilityCol=probabilityCol,
            eps=eps,
        )

    def getMetricName(self) -> str:
        return self.getOrDefault(self.metricName)

    def setMetricName(self, value: str):
        if value not in _SUPPORTED:
            raise ValueError(f"Unsupported metricName {value!r}")
        return self.set(self.metricName, value)

    def isLargerBetter(self) -> bool:
        return self.getMetricName() not in ("hammingLoss", "logLoss")

    # -- core ----------------------------------------------------------------

    def _log_loss(self, dataset: FlowFrame) -> float:
        """Weighted multiclass log loss from probabilityCol (Spark ≥3.1)."""
        prob = dataset.col(self.getOrDefault(self.probabilityCol))
        label = dataset.col(self.getOrDefault(self.labelCol))
        if not isinstance(prob, torch.Tensor) or prob.dim() != 2:
            raise TypeError("probability column must be a vector column for logLoss")
        y = label.reshape(-1).to(torch.int64)
        e = self.getOrDefault(self.eps)
        p = prob.to(torch.float64).gather(1, y.unsqueeze(1)).squeeze(1).clamp_min(e)
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64)
            return float(-(w * p.log()).sum() / w.sum())
        return float(-p.log().mean())

    def _confusion(self, dataset: FlowFrame) -> torch.Tensor:
        pred = dataset.col(self.getOrDefault(self.predictionCol))
        label = dataset.col(self.getOrDefault(self.labelCol))
        if not isinstance(pred, torch.Tensor) or not isinstance(label, torch.Tensor):
            raise TypeError("prediction/label columns must be numeric")
        p = pred.reshape(-1).to(torch.int64)
        y = label.reshape(-1).to(torch.int64)
        k = int(torch.maximum(p.max(), y.max()).item()) + 1 if len(p) else 1
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64)
        else:
            w = torch.ones_like(p, dtype=torch.float64)
        # confusion[label, pred]
        conf = torch.bincount(y * k + p, weights=w, minlength=k * k).reshape(k, k)
        return conf

    def evaluate(self, dataset: FlowFrame) -> float:
        name = self.getMetricName()
        if name not in _SUPPORTED:
            raise ValueError(f"Unsupported metricName {name!r}")
        if name == "logLoss":
            return self._log_loss(dataset)
        conf = self._confusion(dataset)
        total = conf.sum()
        tp = conf.diagonal()
        per_label_actual = conf.sum(dim=1)  # row = true label
        per_label_pred = conf.sum(dim=0)  # col = prediction
        eps = torch.zeros(())
        precision = torch.where(per_label_pred > 0, tp / per_label_pred, tp.new_zeros(tp.shape))
        recall = torch.where(per_label_actual > 0, tp / per_label_actual, tp.new_zeros(tp.shape))
        beta = self.getOrDefault(self.beta)
        b2 = beta * beta

        def fmeasure(p, r):
            denom = b2 * p + r
            return torch.where(denom > 0, (1 + b2) * p * r / denom, p.new_zeros(p.shape))

        if name == "accuracy":
            return float(tp.sum() / total)
        if name == "f1":
            f = fmeasure(precision, recall)
            return float((f * per_label_actual).sum() / total)
        if name == "weightedPrecision":
            return float((precision * per_label_actual).sum() / total)
        if name in ("weightedRecall", "weightedTruePositiveRate"):
            return float((recall * per_label_actual).sum() / total)
        if name == "hammingLoss":
            return float(1.0 - tp.sum() / total)
        if name == "weightedFalsePositiveRate":
            fp = per_label_pred - tp
            neg = total - per_label_actual
            fpr = torch.where(neg > 0, fp / neg, fp.new_zeros(fp.shape))
            return float((fpr * per_label_actual).sum() / total)
        if name == "weightedFMeasure":
            f = fmeasure(precision, recall)
            return float((f * per_label_actual).sum() / total)
        if name == "macroF1":
            f = fmeasure(precision, recall)
            present = per_label_actual > 0
            return float(f[present].mean()) if bool(present.any()) else 0.0
        lbl = int(self.getOrDefault(self.metricLabel))
        if name in ("truePositiveRateByLabel", "recallByLabel"):
            return float(recall[lbl])
        if name == "precisionByLabel":
            return float(precision[lbl])
        if name == "fMeasureByLabel":
            return float(fmeasure(precision, recall)[lbl])
        if name == "falsePositiveRateByLabel":
            fp = per_label_pred[lbl] - tp[lbl]
            neg = total - per_label_actual[lbl]
            return float(fp / neg) if float(neg) > 0 else 0.0
        raise AssertionError(name)


class BinaryClassificationEvaluator(HasLabelCol, HasRawPredictionCol, HasWeightCol, Params):
    """Binary evaluator: areaUnderROC (default) / areaUnderPR.

    Spec: spark:.../ml/evaluation/BinaryClassificationEvaluator.scala — scores
    come from the rawPrediction column (N×2 margins or a 1-D score).
    """

    metricName: Param[str] = Param(
        "undefined",
        "metricName",
        "metric name in evaluation (areaUnderROC|areaUnderPR)",
        TypeConverters.toString,
    )

    def __init__(
        self,
        *,
        rawPredictionCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        metricName: Optional[str] = None,
        weightCol: Optional[str] = None,
    ):
        super().__init__()
        self._setDefault(
            rawPredictionCol="rawPrediction", labelCol="label", metricName="areaUnderROC"
        )
        self._set(
            rawPredictionCol=rawPredictionCol,
            labelCol=labelCol,
            metricName=metricName,
            weightCol=weightCol,
        )

    def getMetricName(self) -> str:
        return self.getOrDefault(self.metricName)

    def isLargerBetter(self) -> bool:
        return True

    def evaluate(self, dataset: FlowFrame) -> float:
        raw = dataset.col(self.getOrDefault(self.rawPredictionCol))
        y = dataset.col(self.getOrDefault(self.labelCol)).reshape(-1).to(torch.float64)
        score = raw[:, 1] if raw.dim() == 2 else raw.reshape(-1)
        score = score.to(torch.float64)
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64)
        else:
            w = torch.ones_like(y)
        order = torch.argsort(score, descending=True)
        y = y[order]
        w = w[order]
        tp = torch.cumsum(w * y, dim=0)
        fp = torch.cumsum(w * (1 - y), dim=0)
        # collapse ties: keep the last point of each distinct score
        s_sorted = score[order]
        last = torch.ones_like(y, dtype=torch.bool)
        last[:-1] = s_sorted[:-1] != s_sorted[1:]
        tp = tp[last]
        fp = fp[last]
        P = float((w * y).sum())
        Nn = float((w * (1 - y)).sum())
        if P == 0 or Nn == 0:
            return 0.0
        tpr = torch.cat([tp.new_zeros(1), tp / P])
        fpr = torch.cat([fp.new_zeros(1), fp / Nn])
        if self.getMetricName() == "areaUnderROC":
            return float(torch.trapz(tpr, fpr))
        # areaUnderPR: precision-recall curve, first point (r0, p=1)
        prec = tp / (tp + fp)
        rec = tp / P
        prec = torch.cat([prec.new_ones(1), prec])
        rec = torch.cat([rec.new_zeros(1), rec])
        return float(torch.trapz(prec, rec))


class RegressionEvaluator(HasLabelCol, HasPredictionCol, HasWeightCol, Params):
    """Spec: spark:.../ml/evaluation/RegressionEvaluator.scala —
    metricName ∈ rmse (default) | mse | mae | r2 | var."""

    metricName: Param[str] = Param(
        "undefined", "metricName", "metric name in evaluation (rmse|mse|r2|mae|var)",
        TypeConverters.toString,
    )

    def __init__(self, *, predictionCol: Optional[str] = None,
                 labelCol: Optional[str] = None, metricName: Optional[str] = None,
                 weightCol: Optional[str] = None):
        super().__init__()
        self._setDefault(predictionCol="prediction", labelCol="label",
                         metricName="rmse")
        self._set(predictionCol=predictionCol, labelCol=labelCol,
                  metricName=metricName, weightCol=weightCol)

    def isLargerBetter(self) -> bool:
        return self.getOrDefault(self.metricName) in ("r2", "var")

    def evaluate(self, dataset: FlowFrame) -> float:
        name = self.getOrDefault(self.metricName)
        if name not in ("rmse", "mse", "mae", "r2", "var"):
            raise ValueError(f"Unsupported metricName {name!r}")
        pred = dataset.col(self.getOrDefault(self.predictionCol)).reshape(-1).to(torch.float64)
        lab = dataset.col(self.getOrDefault(self.labelCol)).reshape(-1).to(torch.float64)
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64)
        else:
            w = torch.ones_like(lab)
        wsum = w.sum()
        err = lab - pred
        mse = float((w * err * err).sum() / wsum)
        if name == "mse":
            return mse
        if name == "rmse":
            return float(mse ** 0.5)
        if name == "mae":
            return float((w * err.abs()).sum() / wsum)
        mean_l = (w * lab).sum() / wsum
        if name == "var":
            # Spark's explainedVariance: SS_reg/n with the LABEL mean
            # (RegressionMetrics.explainedVariance), not the prediction mean
            return float((w * (pred - mean_l) ** 2).sum() / wsum)
        ss_tot = float((w * (lab - mean_l) ** 2).sum())
        ss_res = float((w * err * err).sum())
        return 1.0 - ss_res / ss_tot if ss_tot > 0 else 0.0
