"""Model selection: ParamGridBuilder, CrossValidator, TrainValidationSplit.

Behavioral spec: ``pyspark.ml.tuning`` (upstream ``spark:python/pyspark/ml/tuning.py``):
grid search over param maps with k-fold cross validation (default
``numFolds=3``) or a single train/validation split (``trainRatio=0.75``),
selecting by the evaluator's metric (``isLargerBetter`` respected), then
refitting the best params on the full dataset. ``avgMetrics`` /
``validationMetrics`` are exposed like Spark's.
"""

from __future__ import annotations

import itertools
from typing import Any, Dict, List, Optional

import numpy as np

from flowml.base import Estimator, Model
from flowml.frame import FlowFrame
from flowml.param import Param, Params, TypeConverters


class ParamGridBuilder:
    """Builder for a param grid used in grid search.

    >>> from flowml.classification import LogisticRegression
    >>> lr = LogisticRegression()
    >>> grid = ParamGridBuilder().addGrid(lr.regParam, [0.0, 0.1]).build()
    >>> len(grid)
    2
    """

    def __init__(self):
        self._grid: Dict[Param, List[Any]] = {}

    def addGrid(self, param: Param, values: List[Any]) -> "ParamGridBuilder":
        self._grid[param] = list(values)
        return self

    def baseOn(self, *args) -> "ParamGridBuilder":
        if len(args) == 1 and isinstance(args[0], dict):
            args = list(args[0].items())
        for p, v in args:
            self.addGrid(p, [v])
        return self

    def build(self) -> List[Dict[Param, Any]]:
        keys = list(self._grid.keys())
        combos = itertools.product(*(self._grid[k] for k in keys))
        return [dict(zip(keys, c)) for c in combos]


class _ValidatorParams(Params):
    seed: Param[int] = Param("undefined", "seed", "random seed.", TypeConverters.toInt)
    parallelism: Param[int] = Param(
        "undefined", "parallelism",
        "the number of threads to use when running parallel algorithms (>= 1). "
        "Accepted for Spark API parity; sub-fits run sequentially here — each "
        "fit already saturates the GPU(s), so thread-level fit parallelism "
        "would only interleave kernels.",
        TypeConverters.toInt,
    )

    def __init__(self):
        super().__init__()
        self._setDefault(seed=42, parallelism=1)
        self.estimator: Optional[Estimator] = None
        self.estimatorParamMaps: List[Dict[Param, Any]] = []
        self.evaluator = None

    def setEstimator(self, est: Estimator):
        self.estimator = est
        return self

    def setEstimatorParamMaps(self, maps: List[Dict[Param, Any]]):
        self.estimatorParamMaps = list(maps)
        return self

    def setEvaluator(self, ev):
        self.evaluator = ev
        return self

    def _check(self):
        if self.estimator is None or self.evaluator is None or not self.estimatorParamMaps:
            raise ValueError(
                "estimator, estimatorParamMaps and evaluator must all be set"
            )


class CrossValidator(_ValidatorParams, Estimator):
    numFolds: Param[int] = Param(
        "undefined", "numFolds", "number of folds for cross validation (>= 2)", TypeConverters.toInt
    )
    foldCol: Param[str] = Param(
        "undefined", "foldCol",
        "Param for the column name of user specified fold number. Once this "
        "is specified, CrossValidator won't do random k-fold split. Note "
        "that this column should be integer type with range [0, numFolds).",
        TypeConverters.toString,
    )

    def __init__(self, *, estimator=None, estimatorParamMaps=None, evaluator=None,
                 numFolds: Optional[int] = None, seed: Optional[int] = None,
                 foldCol: Optional[str] = None):
        super().__init__()
        self._setDefault(numFolds=3, foldCol="")
        self._set(numFolds=numFolds, seed=seed, foldCol=foldCol)
        if estimator is not None:
            self.estimator = estimator
        if estimatorParamMaps is not None:
            self.estimatorParamMaps = list(estimatorParamMaps)
        if evaluator is not None:
            self.evaluator = evaluator

    def _fit(self, dataset: FlowFrame) -> "CrossValidatorModel":
        self._check()
        k = self.getOrDefault(self.numFolds)
        seed = self.getOrDefault(self.seed)
        n = dataset.count()
        fold_col = self.getOrDefault(self.foldCol)
        if fold_col:
            # Spark 3.1 foldCol: user-specified integer folds in [0, k)
            fc = dataset.col(fold_col)
            fold_of = fc.detach().cpu().numpy().astype(np.int64).reshape(-1)
            if fold_of.size and (fold_of.min() < 0 or fold_of.max() >= k):
                raise ValueError(
                    f"foldCol {fold_col!r} values must be in [0, {k})")
        else:
            rng = np.random.default_rng(seed)
            fold_of = rng.integers(0, k, n)  # Spark: random fold assignment
        n_maps = len(self.estimatorParamMaps)
        metrics = np.zeros(n_maps, dtype=np.float64)
        for fold in range(k):
            val_mask = fold_of == fold
            train = dataset._take_mask(~val_mask)
            val = dataset._take_mask(val_mask)
            for j, pm in enumerate(self.estimatorParamMaps):
                model = self.estimator.fit(train, pm)
                metrics[j] += self.evaluator.evaluate(model.transform(val))
        metrics /= k
        best_j = int(np.argmax(metrics) if self.evaluator.isLargerBetter() else np.argmin(metrics))
        best_model = self.estimator.fit(dataset, self.estimatorParamMaps[best_j])
        cvm = CrossValidatorModel(bestModel=best_model, avgMetrics=[float(m) for m in metrics])
        self._copyValues(cvm)
        cvm._set_parent(self)
        return cvm


class CrossValidatorModel(Model):
    def __init__(self, bestModel=None, avgMetrics: Optional[List[float]] = None):
        super().__init__()
        self.bestModel = bestModel
        self.avgMetrics = avgMetrics or []

    def _transform(self, dataset):
        return self.bestModel.transform(dataset)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str) -> "CrossValidatorModel":
        from flowml.io import load as _load

        return _load(path)


class TrainValidationSplit(_ValidatorParams, Estimator):
    trainRatio: Param[float] = Param(
        "undefined", "trainRatio", "ratio between train and validation data (0-1)", TypeConverters.toFloat
    )

    def __init__(self, *, estimator=None, estimatorParamMaps=None, evaluator=None,
                 trainRatio: Optional[float] = None, seed: Optional[int] = None):
        super().__init__()
        self._setDefault(trainRatio=0.75)
        self._set(trainRatio=trainRatio, seed=seed)
        if estimator is not None:
            self.estimator = estimator
        if estimatorParamMaps is not None:
            self.estimatorParamMaps = list(estimatorParamMaps)
        if evaluator is not None:
            self.evaluator = evaluator

    def _fit(self, dataset: FlowFrame) -> "TrainValidationSplitModel":
        self._check()
        ratio = self.getOrDefault(self.trainRatio)
        seed = self.getOrDefault(self.seed)
        train, val = dataset.randomSplit([ratio, 1 - ratio], seed=seed)
        metrics = []
        for pm in self.estimatorParamMaps:
            model = self.estimator.fit(train, pm)
            metrics.append(float(self.evaluator.evaluate(model.transform(val))))
        arr = np.asarray(metrics)
        best_j = int(np.argmax(arr) if self.evaluator.isLargerBetter() else np.argmin(arr))
        best_model = self.estimator.fit(dataset, self.estimatorParamMaps[best_j])
        tvsm = TrainValidationSplitModel(bestModel=best_model, validationMetrics=metrics)
        self._copyValues(tvsm)
        tvsm._set_parent(self)
        return tvsm


class TrainValidationSplitModel(Model):
    def __init__(self, bestModel=None, validationMetrics: Optional[List[float]] = None):
        super().__init__()
        self.bestModel = bestModel
        self.validationMetrics = validationMetrics or []

    def _transform(self, dataset):
        return self.bestModel.transform(dataset)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str) -> "TrainValidationSplitModel":
        from flowml.io import load as _load

        return _load(path)
