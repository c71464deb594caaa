"""RandomForestClassifier / RandomForestClassificationModel (+ single tree).

Behavioral spec: ``spark:.../ml/tree/impl/RandomForest.scala`` (SURVEY.md
§2.2): defaults numTrees=20, maxDepth=5, maxBins=32, impurity=gini,
featureSubsetStrategy=auto (√F for classification), subsamplingRate=1.0 with
Poisson bootstrap, minInstancesPerNode=1, minInfoGain=0, maxMemoryInMB=256.
predictRaw sums per-tree leaf class-probability vectors; probability is the
normalized raw vector (majority-prob vote).

The training loop is the level-synchronous replicated grower of
``flowml.classification.tree``: per node-group chunk one K2
histogram pass + one RCCL all_reduce (C2), split scan replicated on every
rank.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from flowml import ops
from flowml.classification.summary import _HasTrainingSummary
from flowml.classification.base import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    _features_tensor,
    _label_tensor,
)
from flowml.classification.tree import (
    FlatTree,
    GrowthConfig,
    TreeEnsemble,
    apply_categorical_codes,
    cat_arity_from_meta,
    find_split_thresholds,
    grow_trees,
)
from flowml.frame import FlowFrame
from flowml.linalg import DenseVector
from flowml.param import HasSeed, HasWeightCol, Param, TypeConverters
from flowml import dist


class _TreeEnsembleParams(HasSeed, HasWeightCol):
    numTrees: Param[int] = Param("undefined", "numTrees", "Number of trees to train (>= 1).", TypeConverters.toInt)
    maxDepth: Param[int] = Param("undefined", "maxDepth", "Maximum depth of the tree (>= 0).", TypeConverters.toInt)
    maxBins: Param[int] = Param("undefined", "maxBins", "Max number of bins for discretizing continuous features.", TypeConverters.toInt)
    minInstancesPerNode: Param[int] = Param("undefined", "minInstancesPerNode", "Minimum number of instances each child must have after split.", TypeConverters.toInt)
    minInfoGain: Param[float] = Param("undefined", "minInfoGain", "Minimum information gain for a split to be considered at a tree node.", TypeConverters.toFloat)
    maxMemoryInMB: Param[int] = Param("undefined", "maxMemoryInMB", "Maximum memory in MB allocated to histogram aggregation.", TypeConverters.toInt)
    impurity: Param[str] = Param("undefined", "impurity", "Criterion used for information gain calculation: entropy, gini (classification) / variance (regression).", TypeConverters.toString)
    featureSubsetStrategy: Param[str] = Param("undefined", "featureSubsetStrategy", "The number of features to consider for splits at each tree node: auto, all, onethird, sqrt, log2, (0.0-1.0], [1-n].", TypeConverters.toString)
    subsamplingRate: Param[float] = Param("undefined", "subsamplingRate", "Fraction of the training data used for learning each decision tree, in range (0, 1].", TypeConverters.toFloat)
    bootstrap: Param[bool] = Param("undefined", "bootstrap", "Whether bootstrap samples are used when building trees.", TypeConverters.toBoolean)
    cacheNodeIds: Param[bool] = Param("undefined", "cacheNodeIds", "If true, cache node IDs for each instance (parity no-op: node ids are always resident tensors here).", TypeConverters.toBoolean)
    checkpointInterval: Param[int] = Param("undefined", "checkpointInterval", "Checkpoint interval (>= 1) or -1 to disable.", TypeConverters.toInt)

    def __init__(self):
        super().__init__()
        self._setDefault(
            numTrees=20,
            maxDepth=5,
            maxBins=32,
            minInstancesPerNode=1,
            minInfoGain=0.0,
            maxMemoryInMB=256,
            impurity="gini",
            featureSubsetStrategy="auto",
            subsamplingRate=1.0,
            bootstrap=True,
            cacheNodeIds=False,
            checkpointInterval=10,
            seed=42,
        )

    def getNumTrees(self) -> int:
        return self.getOrDefault(self.numTrees)

    def getMaxDepth(self) -> int:
        return self.getOrDefault(self.maxDepth)

    def getMaxBins(self) -> int:
        return self.getOrDefault(self.maxBins)

    def getImpurity(self) -> str:
        return self.getOrDefault(self.impurity)


class RandomForestClassifier(_TreeEnsembleParams, ProbabilisticClassifier):
    def __init__(
        self,
        *,
        featuresCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        predictionCol: Optional[str] = None,
        rawPredictionCol: Optional[str] = None,
        probabilityCol: Optional[str] = None,
        numTrees: Optional[int] = None,
        maxDepth: Optional[int] = None,
        maxBins: Optional[int] = None,
        minInstancesPerNode: Optional[int] = None,
        minInfoGain: Optional[float] = None,
        maxMemoryInMB: Optional[int] = None,
        impurity: Optional[str] = None,
        featureSubsetStrategy: Optional[str] = None,
        subsamplingRate: Optional[float] = None,
        bootstrap: Optional[bool] = None,
        seed: Optional[int] = None,
        weightCol: Optional[str] = None,
    ):
        super().__init__()
        self._set(
            featuresCol=featuresCol,
            labelCol=labelCol,
            predictionCol=predictionCol,
            rawPredictionCol=rawPredictionCol,
            probabilityCol=probabilityCol,
            numTrees=numTrees,
            maxDepth=maxDepth,
            maxBins=maxBins,
            minInstancesPerNode=minInstancesPerNode,
            minInfoGain=minInfoGain,
            maxMemoryInMB=maxMemoryInMB,
            impurity=impurity,
            featureSubsetStrategy=featureSubsetStrategy,
            subsamplingRate=subsamplingRate,
            bootstrap=bootstrap,
            seed=seed,
            weightCol=weightCol,
        )

    def setNumTrees(self, v):
        return self.set(self.numTrees, v)

    def setMaxDepth(self, v):
        return self.set(self.maxDepth, v)

    def _fit(self, dataset: FlowFrame) -> "RandomForestClassificationModel":
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.int64)
        dev = X.device
        N, F = X.shape
        kmax = torch.tensor([float(y.max().item()) if N else 0.0], device=dev)
        if dist.is_initialized():
            torch.distributed.all_reduce(kmax, op=torch.distributed.ReduceOp.MAX)
        n_classes = max(int(kmax.item()) + 1, 2)
        import os as _os
        import time as _time

        _timing = _os.environ.get("FLOWML_TIMING", "0") == "1"

        def _tick():
            if _timing and dev.type == "cuda":
                torch.cuda.synchronize()
            return _time.perf_counter()

        t0 = _tick()
        seed = self.getOrDefault(self.seed)
        B = self.getMaxBins()
        thresholds = find_split_thresholds(X, B, seed)
        t1 = _tick()
        codes = ops.bin_codes(X, thresholds)
        cat_arity = cat_arity_from_meta(dataset, self.getFeaturesCol(), F)
        if cat_arity is not None:
            apply_categorical_codes(codes, X, cat_arity)
        t2 = _tick()
        T = self.getNumTrees()
        # BaggedPoint weights are computed IN the K2 kernel from a splitmix64
        # hash of (seed, global row id, tree) — no T×N bag tensor, and the
        # draw is rank-count-invariant (SURVEY.md §7 hard-part 5)
        rate = self.getOrDefault(self.subsamplingRate)
        bootstrap = self.getOrDefault(self.bootstrap)
        hash_bag = None
        if bootstrap or rate < 1.0:
            mode = 2 if bootstrap else 3
            cdf = ops.poisson_cdf_table(rate, "poisson" if bootstrap else "bernoulli")
            row_offset = 0
            if dist.is_initialized() and dist.get_world_size() > 1:
                counts = dist.all_gather_object(int(N))
                row_offset = int(sum(counts[: dist.get_rank()]))
            hash_bag = (mode, cdf.to(dev), (seed * 0x9E3779B9) & 0x7FFFFFFFFFFF, row_offset)
        t3 = _tick()
        row_w = None
        if self.isSet(self.weightCol):
            row_w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float32)
        cfg = GrowthConfig(
            num_trees=T,
            max_depth=self.getMaxDepth(),
            max_bins=B,
            impurity=self.getImpurity(),
            feature_subset=self.getOrDefault(self.featureSubsetStrategy),
            min_instances_per_node=self.getOrDefault(self.minInstancesPerNode),
            min_info_gain=self.getOrDefault(self.minInfoGain),
            max_memory_mb=self.getOrDefault(self.maxMemoryInMB),
            subsampling_rate=self.getOrDefault(self.subsamplingRate),
            bootstrap=self.getOrDefault(self.bootstrap),
            seed=seed,
            is_classification=True,
            n_classes=n_classes,
            cat_arity=cat_arity,
        )
        trees = grow_trees(codes, y, None, thresholds, cfg, row_weights=row_w,
                           hash_bag=hash_bag)
        t4 = _tick()
        model = RandomForestClassificationModel(
            trees=trees, numClasses=n_classes, numFeatures=F
        )
        t5 = _tick()
        from flowml.utils.logging import get_logger

        get_logger().log(
            "rf_fit", trees=T, total_nodes=int(sum(tr.n_nodes for tr in trees)),
            rows=int(N), features=int(F), classes=int(n_classes),
            fit_ms=round((t5 - t0) * 1e3, 2),
        )
        if _timing:
            print("rf_fit timing(ms):", {
                "thresholds": round((t1 - t0) * 1e3, 1),
                "bin": round((t2 - t1) * 1e3, 1),
                "bag": round((t3 - t2) * 1e3, 1),
                "grow": round((t4 - t3) * 1e3, 1),
                "model_build": round((t5 - t4) * 1e3, 1),
            })
        self._copyValues(model)
        model._set_parent(self)
        from flowml.classification.summary import ClassificationTrainingSummary

        model._summary = ClassificationTrainingSummary(
            model, dataset, self.getLabelCol(), self.getPredictionCol()
        )
        return model


class RandomForestClassificationModel(_TreeEnsembleParams, _HasTrainingSummary, ProbabilisticClassificationModel):
    def __init__(
        self,
        trees: Optional[List[FlatTree]] = None,
        numClasses: int = 2,
        numFeatures: int = 0,
    ):
        super().__init__()
        self._trees = trees or []
        self.numClasses = numClasses
        self.numFeatures = numFeatures
        # inference arrays are built lazily: fit() never pays for them
        # (0.7 ms of the 18 ms flagship fit), first transform does
        self._ensemble_cache: Optional[TreeEnsemble] = None

    @property
    def _ensemble(self) -> Optional[TreeEnsemble]:
        if self._ensemble_cache is None and self._trees:
            self._ensemble_cache = TreeEnsemble(self._trees, self.numClasses,
                                                mode="prob_sum")
        return self._ensemble_cache

    @property
    def trees(self) -> List["DecisionTreeClassificationModel"]:
        return [
            DecisionTreeClassificationModel(tree=t, numClasses=self.numClasses, numFeatures=self.numFeatures)
            for t in self._trees
        ]

    @property
    def getNumTreesProperty(self) -> int:
        return len(self._trees)

    @property
    def totalNumNodes(self) -> int:
        return sum(t.n_nodes for t in self._trees)

    @property
    def treeWeights(self) -> List[float]:
        return [1.0] * len(self._trees)

    @property
    def toDebugString(self) -> str:
        from flowml.classification.tree import ensemble_debug_string

        return ensemble_debug_string(
            f"RandomForestClassificationModel: uid={self.uid}, "
            f"numTrees={len(self._trees)}, numClasses={self.numClasses}, "
            f"numFeatures={self.numFeatures}", self._trees)

    @property
    def featureImportances(self) -> DenseVector:
        return DenseVector(self._ensemble.feature_importances(self.numFeatures))

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        return self._ensemble.predict_raw(features)

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        s = raw.sum(dim=1, keepdim=True).clamp_min(1e-30)
        return raw / s

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


class DecisionTreeClassificationModel(_TreeEnsembleParams, _HasTrainingSummary, ProbabilisticClassificationModel):
    """Single decision tree (also the per-tree element of a forest)."""

    def __init__(self, tree: Optional[FlatTree] = None, numClasses: int = 2, numFeatures: int = 0):
        super().__init__()
        self._tree = tree or FlatTree()
        self.numClasses = numClasses
        self.numFeatures = numFeatures
        self._ensemble = TreeEnsemble([self._tree], numClasses, mode="prob_sum")

    @property
    def toDebugString(self) -> str:
        from flowml.classification.tree import tree_debug_string

        return (f"DecisionTreeClassificationModel: uid={self.uid}, "
                f"depth={self.depth}, numNodes={self.numNodes}, "
                f"numClasses={self.numClasses}, "
                f"numFeatures={self.numFeatures}\n"
                + tree_debug_string(self._tree))

    @property
    def depth(self) -> int:
        return self._tree.depth()

    @property
    def numNodes(self) -> int:
        return self._tree.n_nodes

    @property
    def featureImportances(self) -> DenseVector:
        return DenseVector(self._ensemble.feature_importances(self.numFeatures))

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        # single tree raw = leaf class counts (Spark) — we keep normalized
        # probabilities × count to match impurityStats-based raw closely
        return self._ensemble.predict_raw(features)

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        s = raw.sum(dim=1, keepdim=True).clamp_min(1e-30)
        return raw / s


class DecisionTreeClassifier(_TreeEnsembleParams, ProbabilisticClassifier):
    """Single decision tree (spark:.../classification/DecisionTreeClassifier.scala):
    a 1-tree forest with all features considered at every node and no
    bootstrap — defaults maxDepth=5, maxBins=32, impurity=gini."""

    def __init__(
        self,
        *,
        featuresCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        predictionCol: Optional[str] = None,
        rawPredictionCol: Optional[str] = None,
        probabilityCol: Optional[str] = None,
        maxDepth: Optional[int] = None,
        maxBins: Optional[int] = None,
        minInstancesPerNode: Optional[int] = None,
        minInfoGain: Optional[float] = None,
        impurity: Optional[str] = None,
        seed: Optional[int] = None,
        weightCol: Optional[str] = None,
    ):
        super().__init__()
        self._set(
            featuresCol=featuresCol,
            labelCol=labelCol,
            predictionCol=predictionCol,
            rawPredictionCol=rawPredictionCol,
            probabilityCol=probabilityCol,
            maxDepth=maxDepth,
            maxBins=maxBins,
            minInstancesPerNode=minInstancesPerNode,
            minInfoGain=minInfoGain,
            impurity=impurity,
            seed=seed,
            weightCol=weightCol,
        )
        self._set(numTrees=1, bootstrap=False, subsamplingRate=1.0,
                  featureSubsetStrategy="all")

    def _fit(self, dataset: FlowFrame) -> "DecisionTreeClassificationModel":
        rf = RandomForestClassifier()
        self._copyValues(rf)
        forest = rf._fit(dataset)
        model = DecisionTreeClassificationModel(
            tree=forest._trees[0],
            numClasses=forest.numClasses,
            numFeatures=forest.numFeatures,
        )
        self._copyValues(model)
        model._set_parent(self)
        from flowml.classification.summary import ClassificationTrainingSummary

        model._summary = ClassificationTrainingSummary(
            model, dataset, self.getLabelCol(), self.getPredictionCol()
        )
        return model
