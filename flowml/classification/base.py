"""Classifier / ClassificationModel base hierarchy.

Behavioral spec: ``pyspark.ml.classification``'s ``Classifier`` /
``ProbabilisticClassifier`` abstractions: a fitted
classification model's ``transform`` appends, as configured,

* ``rawPredictionCol`` (margins / votes, an N×K tensor column),
* ``probabilityCol``   (normalized class-conditional probabilities),
* ``predictionCol``    (argmax as double),

computing each from the previous (``predictRaw`` → ``raw2probability`` →
argmax), exactly once per transform.
"""

from __future__ import annotations

import torch

from flowml.base import Estimator, Model
from flowml.frame import FlowFrame
from flowml.param import (
    HasFeaturesCol,
    HasLabelCol,
    HasPredictionCol,
    HasProbabilityCol,
    HasRawPredictionCol,
    HasThresholds,
)


class _PredictorParams(HasFeaturesCol, HasLabelCol, HasPredictionCol):
    def __init__(self):
        super().__init__()
        self._setDefault(featuresCol="features", labelCol="label", predictionCol="prediction")


class _ClassifierParams(_PredictorParams, HasRawPredictionCol):
    def __init__(self):
        super().__init__()
        self._setDefault(rawPredictionCol="rawPrediction")


class _ProbabilisticClassifierParams(_ClassifierParams, HasProbabilityCol, HasThresholds):
    def __init__(self):
        super().__init__()
        self._setDefault(probabilityCol="probability")


def _features_tensor(dataset: FlowFrame, col: str) -> torch.Tensor:
    x = dataset.col(col)
    if not isinstance(x, torch.Tensor) or x.dim() != 2:
        raise TypeError(f"features column {col!r} must be a vector column (2-D tensor)")
    return x


def _label_tensor(dataset: FlowFrame, col: str,
                  classification: bool = False) -> torch.Tensor:
    y = dataset.col(col)
    if not isinstance(y, torch.Tensor):
        raise TypeError(f"label column {col!r} must be numeric")
    y = y.reshape(-1)
    if classification and len(y) and bool((y < 0).any()):
        # Spark: "Labels MUST be in [0, maxNumClasses)" — fail clearly
        # instead of crashing in the label indexing later
        raise ValueError(
            f"classification labels in column {col!r} must be non-negative "
            f"integers (found min {float(y.min())})")
    return y


class Classifier(_ClassifierParams, Estimator):
    pass


class ProbabilisticClassifier(_ProbabilisticClassifierParams, Estimator):
    pass


class ClassificationModel(_ClassifierParams, Model):
    """Model producing rawPrediction + prediction columns."""

    numClasses: int = 0
    numFeatures: int = 0

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        """N×K margins. Subclasses implement."""
        raise NotImplementedError

    def raw2prediction(self, raw: torch.Tensor) -> torch.Tensor:
        return raw.argmax(dim=1).to(torch.float64)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = _features_tensor(dataset, self.getFeaturesCol())
        raw = self.predictRaw(x)
        out = dataset
        raw_col = self.getOrDefault(self.rawPredictionCol)
        pred_col = self.getOrDefault(self.predictionCol)
        if raw_col:
            out = out.withColumn(raw_col, raw)
        if pred_col:
            out = out.withColumn(pred_col, self.raw2prediction(raw))
        return out

    def predict(self, features) -> float:
        """Single-row convenience (pyspark Model.predict)."""
        t = _as_row_tensor(features)
        return float(self.raw2prediction(self.predictRaw(t))[0])


class ProbabilisticClassificationModel(_ProbabilisticClassifierParams, ClassificationModel):
    """Adds probabilityCol via raw2probability."""

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        raise NotImplementedError

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = _features_tensor(dataset, self.getFeaturesCol())
        raw = self.predictRaw(x)
        out = dataset
        raw_col = self.getOrDefault(self.rawPredictionCol)
        prob_col = self.getOrDefault(self.probabilityCol)
        pred_col = self.getOrDefault(self.predictionCol)
        if raw_col:
            out = out.withColumn(raw_col, raw)
        prob = self.raw2probability(raw) if (prob_col or pred_col) else None
        if prob_col:
            out = out.withColumn(prob_col, prob)
        if pred_col:
            # Spark predicts from probability for probabilistic models;
            # with thresholds set, predict argmax(p/t) (Spark semantics)
            scored = prob
            if self.isSet(self.thresholds):
                t = torch.as_tensor(self.getThresholds(), dtype=prob.dtype,
                                    device=prob.device)
                if t.numel() != prob.shape[1]:
                    raise ValueError(
                        f"thresholds length {t.numel()} != numClasses {prob.shape[1]}")
                zero = t == 0
                if bool(zero.any()):
                    # Spark: a zero threshold wins outright (at most one)
                    pred = torch.full((prob.shape[0],), float(int(zero.nonzero()[0])),
                                      dtype=torch.float64, device=prob.device)
                    return out.withColumn(pred_col, pred)
                scored = prob / t
            out = out.withColumn(pred_col, scored.argmax(dim=1).to(torch.float64))
        return out

    def predictProbability(self, features) -> torch.Tensor:
        t = _as_row_tensor(features)
        return self.raw2probability(self.predictRaw(t))[0]


def _as_row_tensor(features) -> torch.Tensor:
    from flowml.linalg import Vector

    if isinstance(features, Vector):
        features = features.toArray()
    t = torch.as_tensor(features, dtype=torch.float32)
    if t.dim() == 1:
        t = t.unsqueeze(0)
    return t
