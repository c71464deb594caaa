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


class BinaryClassificationTrainingSummary(ClassificationTrainingSummary):
    """Binary summary with threshold curves
    (spark BinaryLogisticRegressionTrainingSummary): ROC / PR curves and
    precision/recall/F-measure by threshold, computed from the probability
    column over the training predictions."""

    def __init__(self, model, dataset, labelCol: str, predictionCol: str,
                 probabilityCol: str = "probability"):
        super().__init__(model, dataset, labelCol, predictionCol)
        self.probabilityCol = probabilityCol
        self._curve_cache = None

    def _curves(self):
        if self._curve_cache is not None:
            return self._curve_cache
        import torch

        pred = self.predictions
        prob = pred.col(self.probabilityCol)
        score = prob[:, 1] if prob.dim() == 2 else prob.reshape(-1)
        y = pred.col(self.labelCol).reshape(-1).to(torch.float64)
        score = score.to(torch.float64)
        order = torch.argsort(score, descending=True)
        y = y[order]
        s_sorted = score[order]
        tp = torch.cumsum(y, dim=0)
        fp = torch.cumsum(1 - y, dim=0)
        # one curve point per distinct score (last index of each run)
        last = torch.ones_like(y, dtype=torch.bool)
        last[:-1] = s_sorted[:-1] != s_sorted[1:]
        thr = s_sorted[last]
        tp = tp[last]
        fp = fp[last]
        P = float(y.sum())
        Nn = float((1 - y).sum())
        self._curve_cache = (thr, tp, fp, P, Nn)
        return self._curve_cache

    @property
    def roc(self):
        """FlowFrame{FPR, TPR} anchored at (0,0) and (1,1)."""
        import torch

        from flowml.frame import FlowFrame

        thr, tp, fp, P, Nn = self._curves()
        tpr = torch.cat([tp.new_zeros(1), tp / max(P, 1e-300), tp.new_ones(1)])
        fpr = torch.cat([fp.new_zeros(1), fp / max(Nn, 1e-300), fp.new_ones(1)])
        return FlowFrame({"FPR": fpr, "TPR": tpr})

    @property
    def areaUnderROC(self) -> float:
        import torch

        r = self.roc
        return float(torch.trapz(r.col("TPR"), r.col("FPR")))

    @property
    def pr(self):
        """FlowFrame{recall, precision} with the (0, p1) start point."""
        import torch

        from flowml.frame import FlowFrame

        thr, tp, fp, P, Nn = self._curves()
        prec = tp / (tp + fp)
        rec = tp / max(P, 1e-300)
        return FlowFrame({
            "recall": torch.cat([rec.new_zeros(1), rec]),
            "precision": torch.cat([prec[:1], prec]),
        })

    def _by_threshold(self, values):
        from flowml.frame import FlowFrame

        thr, *_ = self._curves()
        return FlowFrame({"threshold": thr, "metric": values})

    @property
    def precisionByThreshold(self):
        thr, tp, fp, P, Nn = self._curves()
        return self._by_threshold(tp / (tp + fp))

    @property
    def recallByThreshold(self):
        thr, tp, fp, P, Nn = self._curves()
        return self._by_threshold(tp / max(P, 1e-300))

    def fMeasureByThreshold(self, beta: float = 1.0):
        thr, tp, fp, P, Nn = self._curves()
        prec = tp / (tp + fp)
        rec = tp / max(P, 1e-300)
        b2 = beta * beta
        f = (1 + b2) * prec * rec / (b2 * prec + rec).clamp_min(1e-300)
        return self._by_threshold(f)


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
