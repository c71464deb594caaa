"""Pipeline / PipelineModel.

Behavioral spec: ``pyspark.ml.Pipeline`` (upstream ``spark:.../ml/Pipeline.scala``
and ``pipeline.py``): ``fit`` iterates stages —
an Estimator is fitted, and (unless it is the last stage) its Model transforms
the dataset for downstream stages; a Transformer passes through and is
collected as-is. The result is a PipelineModel over the fitted/collected
transformer list; ``transform`` folds the stages left to right.
"""

from __future__ import annotations

from typing import Any, List, Optional

from flowml.base import Estimator, Model, Transformer
from flowml.param import Param


class Pipeline(Estimator):
    """A pipeline of featurization + estimator stages.

    >>> import torch
    >>> from flowml import FlowFrame, Pipeline
    >>> from flowml.feature import StringIndexer, VectorAssembler
    >>> from flowml.classification import LogisticRegression
    >>> df = FlowFrame({"proto": ["tcp", "udp", "tcp", "udp"],
    ...                 "x": [1.0, -1.0, 2.0, -2.0],
    ...                 "label": [1.0, 0.0, 1.0, 0.0]})
    >>> pipe = Pipeline(stages=[
    ...     StringIndexer(inputCol="proto", outputCol="p"),
    ...     VectorAssembler(inputCols=["p", "x"], outputCol="features"),
    ...     LogisticRegression(maxIter=10)])
    >>> model = pipe.fit(df)
    >>> model.transform(df)["prediction"].tolist()
    [1.0, 0.0, 1.0, 0.0]
    """

    stages: Param[List[Any]] = Param("undefined", "stages", "a list of pipeline stages")

    def __init__(self, stages: Optional[List[Any]] = None):
        super().__init__()
        if stages is not None:
            self.set(self.stages, list(stages))

    def setStages(self, value: List[Any]) -> "Pipeline":
        return self.set(self.stages, list(value))

    def getStages(self) -> List[Any]:
        return self.getOrDefault(self.stages)

    def _fit(self, dataset) -> "PipelineModel":
        stages = self.getStages()
        for stage in stages:
            if not (isinstance(stage, (Estimator, Transformer))):
                raise TypeError(f"Cannot recognize a pipeline stage of type {type(stage)}.")
        indexOfLastEstimator = -1
        for i, stage in enumerate(stages):
            if isinstance(stage, Estimator):
                indexOfLastEstimator = i
        transformers: List[Transformer] = []
        for i, stage in enumerate(stages):
            if i <= indexOfLastEstimator:
                if isinstance(stage, Estimator):
                    model = stage.fit(dataset)
                    transformers.append(model)
                    if i < indexOfLastEstimator:
                        dataset = model.transform(dataset)
                else:  # Transformer
                    transformers.append(stage)
                    dataset = stage.transform(dataset)
            else:
                transformers.append(stage)
        pm = PipelineModel(transformers)
        self._copyValues(pm)
        pm._set_parent(self)
        return pm

    def copy(self, extra=None) -> "Pipeline":
        that = super().copy(extra)
        if that.isDefined(that.stages):
            stages = [s.copy() for s in that.getStages()]
            that._paramMap[that.getParam("stages")] = stages
        return that

    def write(self):
        from flowml.io import PipelineWriter

        return PipelineWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str) -> "Pipeline":
        from flowml.io import load as _load

        obj = _load(path)
        if not isinstance(obj, cls):
            raise TypeError(f"Loaded object is {type(obj)}, expected {cls}")
        return obj


class PipelineModel(Model):
    """Fitted pipeline: an ordered list of transformers applied in sequence."""

    def __init__(self, stages: List[Transformer]):
        super().__init__()
        self.stages = list(stages)

    def _transform(self, dataset):
        for t in self.stages:
            dataset = t.transform(dataset)
        return dataset

    def copy(self, extra=None) -> "PipelineModel":
        that = super().copy(extra)
        that.stages = [s.copy() for s in self.stages]
        return that

    def write(self):
        from flowml.io import PipelineModelWriter

        return PipelineModelWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str) -> "PipelineModel":
        from flowml.io import load as _load

        obj = _load(path)
        if not isinstance(obj, cls):
            raise TypeError(f"Loaded object is {type(obj)}, expected {cls}")
        return obj
