"""GBTClassifier / GBTClassificationModel (binary gradient-boosted trees).

Behavioral spec: ``spark:.../ml/tree/impl/GradientBoostedTrees.scala``
(SURVEY.md §2.2): binary logistic loss on labels mapped to {-1,+1};
each round fits a variance-impurity regression tree to the loss's negative
gradient (pseudo-residuals) and adds it with ``stepSize`` (default 0.1)
weighting; ``maxIter`` rounds; optional ``validationIndicatorCol`` early
stop; periodic training checkpoints every ``checkpointInterval`` rounds
(SURVEY.md §5.4 — the PeriodicRDDCheckpointer analog is a host-side
checkpoint of the ensemble-so-far + RNG state, resumable via ``initModel``).

Spark semantics preserved: leaf value = mean pseudo-residual of the leaf
(no Newton refinement), margin F(x) = sum(stepSize_i * tree_i(x)),
probability = sigmoid(2 F(x)), prediction by sign. predictRaw = [-F, F]
(spark GBTClassificationModel.predictRaw convention).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from flowml import dist, ops
from flowml.classification.summary import _HasTrainingSummary
from flowml.classification.base import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    _features_tensor,
    _label_tensor,
)
from flowml.classification.forest import _TreeEnsembleParams
from flowml.classification.tree import (
    FlatTree,
    GrowthConfig,
    TreeEnsemble,
    apply_categorical_codes,
    cat_arity_from_meta,
    find_split_thresholds,
    grow_trees,
    poisson_bag,
)
from flowml.frame import FlowFrame
from flowml.linalg import DenseVector
from flowml.param import HasMaxIter, Param, TypeConverters


class _GBTParams(_TreeEnsembleParams, HasMaxIter):
    stepSize: Param[float] = Param(
        "undefined",
        "stepSize",
        "Step size (a.k.a. learning rate) in interval (0, 1] for shrinking "
        "the contribution of each estimator.",
        TypeConverters.toFloat,
    )
    lossType: Param[str] = Param(
        "undefined", "lossType", "Loss function which GBT tries to minimize: logistic.", TypeConverters.toString
    )
    validationIndicatorCol: Param[str] = Param(
        "undefined",
        "validationIndicatorCol",
        "Name of the column that indicates whether each row is for training "
        "or for validation (early stopping).",
        TypeConverters.toString,
    )
    validationTol: Param[float] = Param(
        "undefined", "validationTol", "Threshold for stopping fitting based on validation loss improvement.", TypeConverters.toFloat
    )

    def __init__(self):
        super().__init__()
        self._setDefault(
            maxIter=20,
            stepSize=0.1,
            lossType="logistic",
            maxDepth=5,
            maxBins=32,
            minInstancesPerNode=1,
            minInfoGain=0.0,
            maxMemoryInMB=256,
            subsamplingRate=1.0,
            featureSubsetStrategy="all",
            validationTol=0.01,
            impurity="variance",
            numTrees=1,
            bootstrap=False,
        )

    def getStepSize(self) -> float:
        return self.getOrDefault(self.stepSize)


class GBTClassifier(_GBTParams, ProbabilisticClassifier):
    def __init__(
        self,
        *,
        featuresCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        predictionCol: Optional[str] = None,
        rawPredictionCol: Optional[str] = None,
        probabilityCol: Optional[str] = None,
        maxIter: Optional[int] = None,
        stepSize: Optional[float] = None,
        maxDepth: Optional[int] = None,
        maxBins: Optional[int] = None,
        minInstancesPerNode: Optional[int] = None,
        minInfoGain: Optional[float] = None,
        maxMemoryInMB: Optional[int] = None,
        subsamplingRate: Optional[float] = None,
        featureSubsetStrategy: Optional[str] = None,
        lossType: Optional[str] = None,
        validationIndicatorCol: Optional[str] = None,
        validationTol: Optional[float] = None,
        checkpointInterval: Optional[int] = None,
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
            maxIter=maxIter,
            stepSize=stepSize,
            maxDepth=maxDepth,
            maxBins=maxBins,
            minInstancesPerNode=minInstancesPerNode,
            minInfoGain=minInfoGain,
            maxMemoryInMB=maxMemoryInMB,
            subsamplingRate=subsamplingRate,
            featureSubsetStrategy=featureSubsetStrategy,
            lossType=lossType,
            validationIndicatorCol=validationIndicatorCol,
            validationTol=validationTol,
            checkpointInterval=checkpointInterval,
            seed=seed,
            weightCol=weightCol,
        )
        self._checkpoint_dir: Optional[str] = None
        self._init_model: Optional["GBTClassificationModel"] = None

    def setCheckpointDir(self, path: str) -> "GBTClassifier":
        """Directory for periodic training checkpoints."""
        self._checkpoint_dir = path
        return self

    def setInitModel(self, model: "GBTClassificationModel") -> "GBTClassifier":
        """Resume boosting from an existing ensemble (checkpoint-restart)."""
        self._init_model = model
        return self

    def _fit(self, dataset: FlowFrame) -> "GBTClassificationModel":
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y01 = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.float32)
        dev = X.device
        N, F = X.shape
        if bool((y01.max() > 1).item()) if N else False:
            raise ValueError("GBTClassifier only supports binary labels {0, 1}")
        ysign = 2.0 * y01 - 1.0  # {-1,+1}

        val_mask = None
        if self.isSet(self.validationIndicatorCol):
            vm = dataset.col(self.getOrDefault(self.validationIndicatorCol))
            val_mask = vm.to(torch.bool).reshape(-1)

        seed = self.getOrDefault(self.seed)
        B = self.getMaxBins()
        thresholds = find_split_thresholds(X, B, seed)
        codes = ops.bin_codes(X, thresholds)
        cat_arity = cat_arity_from_meta(dataset, self.getFeaturesCol(), F)
        if cat_arity is not None:
            apply_categorical_codes(codes, X, cat_arity)

        step = self.getStepSize()
        max_iter = self.getOrDefault(self.maxIter)
        ckpt_int = self.getOrDefault(self.checkpointInterval)

        trees: List[FlatTree] = []
        weights: List[float] = []
        margin = torch.zeros(N, dtype=torch.float32, device=dev)
        start_round = 0
        if self._init_model is not None:
            trees = list(self._init_model._trees)
            weights = list(self._init_model._tree_weights)
            margin = self._init_model._margin(X)
            start_round = len(trees)

        cfg = GrowthConfig(
            num_trees=1,
            max_depth=self.getMaxDepth(),
            max_bins=B,
            impurity="variance",
            feature_subset=self.getOrDefault(self.featureSubsetStrategy),
            min_instances_per_node=self.getOrDefault(self.minInstancesPerNode),
            min_info_gain=self.getOrDefault(self.minInfoGain),
            max_memory_mb=self.getOrDefault(self.maxMemoryInMB),
            seed=seed,
            is_classification=False,
            n_classes=2,
            cat_arity=cat_arity,
        )
        subsample = self.getOrDefault(self.subsamplingRate)

        best_val = float("inf")
        val_tol = self.getOrDefault(self.validationTol)

        train_sel = (~val_mask) if val_mask is not None else None
        inst_w = None
        if self.isSet(self.weightCol):
            # instance weights enter exactly like bag weights: the histogram
            # stats become (Σ w·r, Σ w·r², Σ w) so leaf values are weighted
            # mean residuals (Spark 3.0 GBT weightCol semantics)
            inst_w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float32)

        import os as _os
        import time as _time

        _timing = _os.environ.get("FLOWML_TIMING", "0") == "1"
        _tmr = {}

        def _tick():
            if _timing and dev.type == "cuda":
                torch.cuda.synchronize()
            return _time.perf_counter()

        def _acc(k, t0):
            if _timing:
                _tmr[k] = _tmr.get(k, 0.0) + (_tick() - t0)

        for rnd in range(start_round, max_iter):
            _t0 = _tick()
            # pseudo-residual: -dL/dF for L = log(1+exp(-2 y F)) → 2y/(1+exp(2yF))
            resid = 2.0 * ysign / (1.0 + torch.exp(2.0 * ysign * margin))
            cfg.seed = (seed + 1_000_003 * rnd) & 0x7FFFFFFF
            bag = poisson_bag(N, 1, subsample, False, cfg.seed, dev) if subsample < 1.0 else None
            if train_sel is not None:
                sel = train_sel.to(torch.float32).unsqueeze(0)  # (1, N)
                bag = sel if bag is None else bag * sel
            if inst_w is not None:
                wrow = inst_w.unsqueeze(0)
                bag = wrow.clone() if bag is None else bag * wrow
            _acc("resid", _t0)
            _t1 = _tick()
            tree_list, row_vals = grow_trees(codes, resid, bag, thresholds, cfg,
                                             return_row_values=True)
            tree = tree_list[0]
            trees.append(tree)
            _acc("grow", _t1)
            from flowml.utils.logging import get_logger

            get_logger().log("gbt_round", round=rnd, nodes=tree.n_nodes)
            weights.append(1.0 if rnd == 0 else step)  # Spark: first tree unshrunk
            _t3 = _tick()
            # margin update by leaf-value gather over the settled node ids —
            # bit-identical to re-predicting the new tree (strict-< binning)
            # without the K8 pass over raw X
            margin = margin + weights[-1] * row_vals[0]
            _acc("margin", _t3)
            if val_mask is not None:
                vls = torch.log1p(torch.exp(-2.0 * ysign[val_mask] * margin[val_mask]))
                if inst_w is not None:
                    wv = inst_w[val_mask]
                    buf = torch.tensor([float((vls * wv).sum()), float(wv.sum())], device=dev)
                else:
                    buf = torch.tensor([float(vls.sum()), float(len(vls))], device=dev)
                dist.all_reduce_(buf)
                vloss = float(buf[0] / buf[1])
                if vloss < best_val - val_tol * max(best_val, 1e-12):
                    best_val = vloss
                elif vloss >= best_val:
                    break
            if (
                self._checkpoint_dir
                and ckpt_int > 0
                and (rnd + 1) % ckpt_int == 0
                and dist.get_rank() == 0
            ):
                self._write_checkpoint(trees, weights, rnd + 1, F)

        if _timing and _tmr:
            print("gbt rounds timing:", {k: round(v * 1000, 1) for k, v in _tmr.items()})
        model = GBTClassificationModel(
            trees=trees, treeWeights=weights, numFeatures=F
        )
        self._copyValues(model)
        model._set_parent(self)
        from flowml.classification.summary import ClassificationTrainingSummary

        model._summary = ClassificationTrainingSummary(
            model, dataset, self.getLabelCol(), self.getPredictionCol()
        )
        return model

    def _write_checkpoint(self, trees, weights, rounds, num_features):
        from flowml.utils.checkpoint import save_gbt_checkpoint

        save_gbt_checkpoint(self._checkpoint_dir, trees, weights, rounds, num_features)


class GBTClassificationModel(_GBTParams, _HasTrainingSummary, ProbabilisticClassificationModel):
    def __init__(
        self,
        trees: Optional[List[FlatTree]] = None,
        treeWeights: Optional[List[float]] = None,
        numFeatures: int = 0,
    ):
        super().__init__()
        self._trees = trees or []
        self._tree_weights = list(treeWeights) if treeWeights else [1.0] * len(self._trees)
        self.numClasses = 2
        self.numFeatures = numFeatures
        self._ensemble = TreeEnsemble(self._trees, 2, mode="scalar")
        self._ensemble.set_tree_weights(self._tree_weights)

    @property
    def trees(self):
        from flowml.classification.forest import DecisionTreeClassificationModel

        return [
            DecisionTreeClassificationModel(tree=t, numClasses=2, numFeatures=self.numFeatures)
            for t in self._trees
        ]

    @property
    def treeWeights(self) -> List[float]:
        return list(self._tree_weights)

    @property
    def getNumTreesProperty(self) -> int:
        return len(self._trees)

    @property
    def totalNumNodes(self) -> int:
        return sum(t.n_nodes for t in self._trees)

    @property
    def featureImportances(self) -> DenseVector:
        return DenseVector(self._ensemble.feature_importances(self.numFeatures))

    def _margin(self, X: torch.Tensor) -> torch.Tensor:
        return self._ensemble.predict_raw(X)[:, 0]

    def evaluateEachIteration(self, dataset, loss: str = "logistic"):
        """Per-boosting-stage training loss (spark
        GBTClassificationModel.evaluateEachIteration): cumulative margin
        after each tree, mean logistic loss log(1 + exp(-2 y' F_m))."""
        if loss != "logistic":
            raise ValueError("only loss='logistic' is supported")
        from flowml.classification.base import _features_tensor, _label_tensor
        from flowml.classification.tree import TreeEnsemble

        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.float32)
        ysign = (2.0 * y - 1.0).to(X.device)
        margin = torch.zeros(len(y), dtype=torch.float32, device=X.device)
        out = []
        for tree, w in zip(self._trees, self._tree_weights):
            single = TreeEnsemble([tree], 2, mode="scalar").to(X.device)
            margin = margin + w * single.predict_raw(X)[:, 0]
            out.append(float(torch.log1p(
                torch.exp(-2.0 * ysign * margin)).mean()))
        return out

    @property
    def toDebugString(self) -> str:
        from flowml.classification.tree import ensemble_debug_string

        return ensemble_debug_string(
            f"GBTClassificationModel: uid={self.uid}, "
            f"numTrees={len(self._trees)}, numFeatures={self.numFeatures}",
            self._trees, self._tree_weights)

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        m = self._margin(features)
        return torch.stack([-m, m], dim=1)

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        # Spark LogLoss.computeProbability: p1 = 1 / (1 + exp(-2 margin))
        p1 = torch.sigmoid(2.0 * raw[:, 1])
        return torch.stack([1 - p1, p1], dim=1)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)
