"""Regression tree ensembles: DecisionTree / RandomForest / GBT regressors.

Behavioral spec: ``pyspark.ml.regression``'s tree models
(``spark:.../ml/regression/RandomForestRegressor.scala``,
``GBTRegressor.scala``, ``DecisionTreeRegressor.scala``): variance-impurity
trees over the same binned-histogram machinery as the classifiers
(``flowml/classification/tree.py`` — SURVEY.md §2.4 K2b/K3b), leaf value =
(weighted) mean target, RF predicts the mean over trees, GBT the
shrinkage-weighted sum of per-round trees fit to squared-loss residuals.

Beyond the reference app's scope (it is a classifier app) — included so the
tree engine serves both MLlib tree families.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from flowml import dist, ops
from flowml.base import Estimator, Model
from flowml.classification.base import _features_tensor, _label_tensor, _PredictorParams
from flowml.classification.forest import _TreeEnsembleParams
from flowml.classification.tree import (
    FlatTree,
    GrowthConfig,
    TreeEnsemble,
    find_split_thresholds,
    grow_trees,
    poisson_bag,
)
from flowml.frame import FlowFrame
from flowml.param import HasMaxIter, Param, TypeConverters


class _RegressorIOMixin:
    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        obj = _load(path)
        if not isinstance(obj, cls):
            raise TypeError(f"Loaded object is {type(obj)}, expected {cls}")
        return obj


class RegressionModel(_PredictorParams, Model):
    """Base: transform appends predictionCol (Spark PredictionModel)."""

    numFeatures: int = 0

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:  # N
        raise NotImplementedError

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = _features_tensor(dataset, self.getFeaturesCol())
        pred = self.predictRaw(x).to(torch.float64)
        return dataset.withColumn(self.getOrDefault(self.predictionCol), pred)

    def predict(self, features) -> float:
        from flowml.classification.base import _as_row_tensor

        return float(self.predictRaw(_as_row_tensor(features))[0])


def _fit_reg_forest(est, dataset: FlowFrame, num_trees: int, use_hash_bag: bool):
    """Shared RF/DT regression fit: variance trees over the shared grower."""
    X = _features_tensor(dataset, est.getFeaturesCol()).to(torch.float32)
    y = _label_tensor(dataset, est.getLabelCol()).to(torch.float32)
    dev = X.device
    N, F = X.shape
    seed = est.getOrDefault(est.seed)
    B = est.getMaxBins()
    thresholds = find_split_thresholds(X, B, seed)
    codes = ops.bin_codes(X, thresholds)
    hash_bag = None
    if use_hash_bag:
        rate = est.getOrDefault(est.subsamplingRate)
        bootstrap = est.getOrDefault(est.bootstrap)
        if bootstrap or rate < 1.0:
            mode = 2 if bootstrap else 3
            cdf = ops.poisson_cdf_table(rate, "poisson" if bootstrap else "bernoulli")
            row_offset = 0
            if dist.is_initialized() and dist.get_world_size() > 1:
                counts = dist.all_gather_object(int(N))
                row_offset = int(sum(counts[: dist.get_rank()]))
            hash_bag = (mode, cdf.to(dev), (seed * 0x9E3779B9) & 0x7FFFFFFFFFFF, row_offset)
    row_w = None
    if est.isSet(est.weightCol):
        row_w = dataset.col(est.getWeightCol()).reshape(-1).to(torch.float32)
    cfg = GrowthConfig(
        num_trees=num_trees,
        max_depth=est.getMaxDepth(),
        max_bins=B,
        impurity="variance",
        feature_subset=est.getOrDefault(est.featureSubsetStrategy),
        min_instances_per_node=est.getOrDefault(est.minInstancesPerNode),
        min_info_gain=est.getOrDefault(est.minInfoGain),
        max_memory_mb=est.getOrDefault(est.maxMemoryInMB),
        subsampling_rate=est.getOrDefault(est.subsamplingRate),
        bootstrap=est.getOrDefault(est.bootstrap),
        seed=seed,
        is_classification=False,
        n_classes=2,
    )
    trees = grow_trees(codes, y, None, thresholds, cfg, row_weights=row_w,
                       hash_bag=hash_bag)
    return trees, F


class RandomForestRegressor(_TreeEnsembleParams, _PredictorParams, Estimator, _RegressorIOMixin):
    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(impurity="variance", featureSubsetStrategy="auto")
        self._set(**kwargs)

    def _fit(self, dataset: FlowFrame) -> "RandomForestRegressionModel":
        trees, F = _fit_reg_forest(self, dataset, self.getNumTrees(), True)
        model = RandomForestRegressionModel(trees=trees, numFeatures=F)
        self._copyValues(model)
        model._set_parent(self)
        return model


class RandomForestRegressionModel(_TreeEnsembleParams, RegressionModel, _RegressorIOMixin):
    def __init__(self, trees: Optional[List[FlatTree]] = None, numFeatures: int = 0):
        super().__init__()
        self._trees = trees or []
        self.numFeatures = numFeatures
        self._ensemble = TreeEnsemble(self._trees, 1, mode="scalar") if trees else None

    @property
    def trees(self):
        return list(self._trees)

    def getNumTrees(self) -> int:  # Spark exposes model.getNumTrees
        return len(self._trees)

    @property
    def featureImportances(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._ensemble.feature_importances(self.numFeatures))

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        raw = self._ensemble.predict_raw(features)[:, 0]
        return raw / max(len(self._trees), 1)


class DecisionTreeRegressor(_TreeEnsembleParams, _PredictorParams, Estimator, _RegressorIOMixin):
    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(impurity="variance", numTrees=1, bootstrap=False,
                         featureSubsetStrategy="all")
        self._set(**kwargs)

    def _fit(self, dataset: FlowFrame) -> "DecisionTreeRegressionModel":
        trees, F = _fit_reg_forest(self, dataset, 1, False)
        model = DecisionTreeRegressionModel(tree=trees[0], numFeatures=F)
        self._copyValues(model)
        model._set_parent(self)
        return model


class DecisionTreeRegressionModel(_TreeEnsembleParams, RegressionModel, _RegressorIOMixin):
    def __init__(self, tree: Optional[FlatTree] = None, numFeatures: int = 0):
        super().__init__()
        self._trees = [tree] if tree is not None else []
        self.numFeatures = numFeatures
        self._ensemble = TreeEnsemble(self._trees, 1, mode="scalar") if tree is not None else None

    @property
    def depth(self) -> int:
        return self._trees[0].depth() if self._trees else 0

    @property
    def numNodes(self) -> int:
        return self._trees[0].n_nodes if self._trees else 0

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        return self._ensemble.predict_raw(features)[:, 0]


class GBTRegressor(_TreeEnsembleParams, _PredictorParams, HasMaxIter, Estimator, _RegressorIOMixin):
    """Squared-loss gradient boosting (spark GBTRegressor, lossType=squared)."""

    stepSize: Param[float] = Param(
        "undefined", "stepSize",
        "Step size (a.k.a. learning rate) in interval (0, 1].",
        TypeConverters.toFloat,
    )
    lossType: Param[str] = Param(
        "undefined", "lossType", "Loss function GBT tries to minimize: squared",
        TypeConverters.toString,
    )

    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(impurity="variance", maxIter=20, stepSize=0.1,
                        lossType="squared", featureSubsetStrategy="all",
                        bootstrap=False)
        self._set(**kwargs)

    def _fit(self, dataset: FlowFrame) -> "GBTRegressionModel":
        if self.getOrDefault(self.lossType) != "squared":
            raise ValueError("GBTRegressor supports lossType='squared'")
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol()).to(torch.float32)
        dev = X.device
        N, F = X.shape
        seed = self.getOrDefault(self.seed)
        B = self.getMaxBins()
        thresholds = find_split_thresholds(X, B, seed)
        codes = ops.bin_codes(X, thresholds)
        step = self.getOrDefault(self.stepSize)
        max_iter = self.getOrDefault(self.maxIter)
        subsample = self.getOrDefault(self.subsamplingRate)
        trees: List[FlatTree] = []
        weights: List[float] = []
        margin = torch.zeros(N, dtype=torch.float32, device=dev)
        for rnd in range(max_iter):
            resid = y - margin  # -∂/∂F of ½(y−F)²
            cfg = GrowthConfig(
                num_trees=1,
                max_depth=self.getMaxDepth(),
                max_bins=B,
                impurity="variance",
                feature_subset=self.getOrDefault(self.featureSubsetStrategy),
                min_instances_per_node=self.getOrDefault(self.minInstancesPerNode),
                min_info_gain=self.getOrDefault(self.minInfoGain),
                max_memory_mb=self.getOrDefault(self.maxMemoryInMB),
                seed=(seed + 1_000_003 * rnd) & 0x7FFFFFFF,
                is_classification=False,
                n_classes=2,
            )
            bag = poisson_bag(N, 1, subsample, False, cfg.seed, dev) if subsample < 1.0 else None
            tree = grow_trees(codes, resid, bag, thresholds, cfg)[0]
            trees.append(tree)
            weights.append(1.0 if rnd == 0 else step)
            te = TreeEnsemble([tree], 1, mode="scalar")
            margin = margin + weights[-1] * te.predict_raw(X)[:, 0]
        model = GBTRegressionModel(trees=trees, treeWeights=weights, numFeatures=F)
        self._copyValues(model)
        model._set_parent(self)
        return model


class GBTRegressionModel(_TreeEnsembleParams, RegressionModel, _RegressorIOMixin):
    def __init__(self, trees: Optional[List[FlatTree]] = None,
                 treeWeights: Optional[List[float]] = None, numFeatures: int = 0):
        super().__init__()
        self._trees = trees or []
        self.treeWeights = list(treeWeights or [])
        self.numFeatures = numFeatures
        if trees:
            self._ensemble = TreeEnsemble(self._trees, 1, mode="scalar")
            self._ensemble.set_tree_weights(self.treeWeights)
        else:
            self._ensemble = None

    @property
    def trees(self):
        return list(self._trees)

    def getNumTrees(self) -> int:
        return len(self._trees)

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        return self._ensemble.predict_raw(features)[:, 0]
