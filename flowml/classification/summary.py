"""Training summaries (Spark ≥3.1 parity).

Behavioral spec: ``pyspark.ml.classification``'s ``*TrainingSummary``
hierarchy (``LogisticRegressionTrainingSummary``,
``RandomForestClassificationTrainingSummary``, ``GBTClassificationTrainingSummary``,
``MultilayerPerceptronClassificationTrainingSummary``): after ``fit``, the
model exposes ``summary`` with the training-set ``predictions`` frame plus
aggregate metrics. Spark computes these lazily over its DataFrame; here the
summary holds the (device-resident) training frame and transforms on first
access, then delegates each metric to ``MulticlassClassificationEvaluator``.

Loaded-from-disk models have no summary (``hasSummary`` False, ``summary``
raises) — same contract as Spark.
"""

from __future__ import annotations

from typing import List


class ClassificationTrainingSummary:
    """Lazy training-set metrics over the fitted model's predictions."""

    def __init__(self, model, dataset, labelCol: str, predictionCol: str):
        self._model = model
        self._dataset = dataset
        self.labelCol = labelCol
        self.predictionCol = predictionCol
        self._pred_frame = None

    @property
    def predictions(self):
        if self._pred_frame is None:
            self._pred_frame = self._model.transform(self._dataset)
        return self._pred_frame

    def _eval(self, name: str, **kw) -> float:
        from flowml.evaluation import MulticlassClassificationEvaluator

        ev = MulticlassClassificationEvaluator(
            metricName=name, labelCol=self.labelCol, predictionCol=self.predictionCol, **kw
        )
        return ev.evaluate(self.predictions)

    @property
    def labels(self) -> List[float]:
        import torch

        y = self.predictions.col(self.labelCol).reshape(-1)
        return sorted(float(v) for v in torch.unique(y).cpu())

    @property
    def accuracy(self) -> float:
        return self._eval("accuracy")

    @property
    def weightedPrecision(self) -> float:
        return self._eval("weightedPrecision")

    @property
    def weightedRecall(self) -> float:
        return self._eval("weightedRecall")

    @property
    def weightedTruePositiveRate(self) -> float:
        return self._eval("weightedTruePositiveRate")

    @property
    def weightedFalsePositiveRate(self) -> float:
        return self._eval("weightedFalsePositiveRate")

    def weightedFMeasure(self, beta: float = 1.0) -> float:
        return self._eval("weightedFMeasure", beta=beta)

    @property
    def precisionByLabel(self) -> List[float]:
        return [self._eval("precisionByLabel", metricLabel=l) for l in self.labels]

    @property
    def recallByLabel(self) -> List[float]:
        return [self._eval("recallByLabel", metricLabel=l) for l in self.labels]

    def fMeasureByLabel(self, beta: float = 1.0) -> List[float]:
        return [self._eval("fMeasureByLabel", metricLabel=l, beta=beta) for l in self.labels]

    @property
    def truePositiveRateByLabel(self) -> List[float]:
        return [self._eval("truePositiveRateByLabel", metricLabel=l) for l in self.labels]

    @property
    def falsePositiveRateByLabel(self) -> List[float]:
        return [self._eval("falsePositiveRateByLabel", metricLabel=l) for l in self.labels]


class _HasTrainingSummary:
    """Mixin giving models Spark's summary/hasSummary contract."""

    _summary = None

    def hasSummary(self) -> bool:
        return self._summary is not None

    @property
    def summary(self) -> ClassificationTrainingSummary:
        if self._summary is None:
            raise RuntimeError(
                f"No training summary available for this {type(self).__name__}"
            )
        return self._summary
