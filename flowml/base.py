"""Estimator / Transformer / Model abstractions.

Behavioral spec: ``pyspark.ml.base`` (upstream ``spark:python/pyspark/ml/base.py``;
SURVEY.md §2.2 row 2): ``Estimator.fit(df) -> Model``,
``Transformer.transform(df) -> df'``; a ``Model`` is a ``Transformer`` holding
learned state plus the parent estimator's params.
"""

from __future__ import annotations

from abc import ABCMeta, abstractmethod
from typing import Any, Dict, Optional

from flowml.param import Param, Params


class _DefaultParamsWritable:
    """Default Spark-format persistence for parameter-only stages
    (DefaultParamsWritable/-Readable): unfitted estimators and stateless
    transformers save metadata-only dirs and reload by class + params.
    Fitted Models override nothing — their data writers take precedence via
    the same GenericWriter dispatch."""

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


class Estimator(_DefaultParamsWritable, Params, metaclass=ABCMeta):
    """Abstract estimator that fits a model on a FlowFrame."""

    @abstractmethod
    def _fit(self, dataset):
        raise NotImplementedError()

    def fit(self, dataset, params: Optional[Dict[Param, Any]] = None):
        """Fit a model. ``params`` overrides embedded params for this call."""
        if params:
            return self.copy(params)._fit(dataset)
        return self._fit(dataset)


class Transformer(_DefaultParamsWritable, Params, metaclass=ABCMeta):
    """Abstract transformer mapping a FlowFrame to a FlowFrame."""

    @abstractmethod
    def _transform(self, dataset):
        raise NotImplementedError()

    def transform(self, dataset, params: Optional[Dict[Param, Any]] = None):
        if params:
            return self.copy(params)._transform(dataset)
        return self._transform(dataset)


class Model(Transformer, metaclass=ABCMeta):
    """Fitted model: a Transformer with learned state.

    ``parent`` points at the Estimator that produced it (pyspark convention).
    """

    parent: Optional[Estimator] = None

    def _set_parent(self, parent: Estimator) -> "Model":
        self.parent = parent
        return self

    def hasParent(self) -> bool:
        return self.parent is not None
