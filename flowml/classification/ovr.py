"""OneVsRest (spark:.../ml/classification/OneVsRest.scala).

Reduces multiclass to K one-vs-all binary problems of any base binary
classifier; prediction = argmax of the per-class positive raw scores.
Sub-fits run sequentially on the resident shard (each is itself
data-parallel across ranks, so there is nothing to gain from nesting).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from flowml.base import Estimator, Model
from flowml.classification.base import (
    ClassificationModel,
    _ClassifierParams,
    _features_tensor,
    _label_tensor,
)
from flowml.frame import FlowFrame


class OneVsRest(_ClassifierParams, Estimator):
    def __init__(self, *, classifier: Optional[Estimator] = None, **kwargs):
        super().__init__()
        self.classifier = classifier
        self._set(**kwargs)

    def setClassifier(self, clf: Estimator) -> "OneVsRest":
        self.classifier = clf
        return self

    def getClassifier(self) -> Optional[Estimator]:
        return self.classifier

    def _fit(self, dataset: FlowFrame) -> "OneVsRestModel":
        if self.classifier is None:
            raise ValueError("OneVsRest requires a classifier to be set")
        y = _label_tensor(dataset, self.getLabelCol(), classification=True)
        num_classes = int(y.max().item()) + 1 if len(y) else 2
        label_col = self.getLabelCol()
        models: List[ClassificationModel] = []
        for k in range(num_classes):
            binary = (y == k).to(torch.float64)
            sub = dataset.withColumn("_ovr_label", binary)
            clf = self.classifier.copy()
            clf.set(clf.getParam("labelCol"), "_ovr_label")
            clf.set(clf.getParam("featuresCol"), self.getFeaturesCol())
            models.append(clf.fit(sub))
        model = OneVsRestModel(models=models, numClasses=num_classes)
        self._copyValues(model)
        model._set_parent(self)
        return model


class OneVsRestModel(_ClassifierParams, Model):
    def __init__(self, models: Optional[List[ClassificationModel]] = None,
                 numClasses: int = 0):
        super().__init__()
        self.models = models or []
        self.numClasses = numClasses

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = _features_tensor(dataset, self.getFeaturesCol())
        scores = []
        for m in self.models:
            raw = m.predictRaw(x)
            # positive-class score: column 1 of the binary raw output
            scores.append(raw[:, 1] if raw.shape[1] >= 2 else raw[:, 0])
        raw = torch.stack(scores, dim=1)
        out = dataset
        raw_col = self.getOrDefault(self.rawPredictionCol)
        pred_col = self.getOrDefault(self.predictionCol)
        if raw_col:
            out = out.withColumn(raw_col, raw)
        if pred_col:
            out = out.withColumn(pred_col, raw.argmax(dim=1).to(torch.float64))
        return out

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str) -> "OneVsRestModel":
        from flowml.io import load as _load

        return _load(path)
