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
    apply_categorical_codes,
    cat_arity_from_meta,
    find_split_thresholds,
    grow_trees,
    poisson_bag,
)
from flowml.frame import FlowFrame
from flowml.param import HasMaxIter, Param, TypeConverters


def np_asarray_f32(v):
    import numpy as _np

    return _np.asarray(v, dtype=_np.float32)


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
    cat_arity = cat_arity_from_meta(dataset, est.getFeaturesCol(), F)
    if cat_arity is not None:
        apply_categorical_codes(codes, X, cat_arity)
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
        cat_arity=cat_arity,
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
    def toDebugString(self) -> str:
        from flowml.classification.tree import ensemble_debug_string

        return ensemble_debug_string(
            f"RandomForestRegressionModel: uid={self.uid}, "
            f"numTrees={len(self._trees)}, numFeatures={self.numFeatures}",
            self._trees)

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

    @property
    def toDebugString(self) -> str:
        from flowml.classification.tree import tree_debug_string

        return (f"DecisionTreeRegressionModel: uid={self.uid}, "
                f"depth={self.depth}, numNodes={self.numNodes}, "
                f"numFeatures={self.numFeatures}\n"
                + (tree_debug_string(self._trees[0]) if self._trees else ""))

    @property
    def featureImportances(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._ensemble.feature_importances(self.numFeatures))

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
        cat_arity = cat_arity_from_meta(dataset, self.getFeaturesCol(), F)
        if cat_arity is not None:
            apply_categorical_codes(codes, X, cat_arity)
        step = self.getOrDefault(self.stepSize)
        max_iter = self.getOrDefault(self.maxIter)
        subsample = self.getOrDefault(self.subsamplingRate)
        inst_w = None
        if self.isSet(self.weightCol):
            inst_w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float32)
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
                cat_arity=cat_arity,
            )
            bag = poisson_bag(N, 1, subsample, False, cfg.seed, dev) if subsample < 1.0 else None
            if inst_w is not None:
                wrow = inst_w.unsqueeze(0)
                bag = wrow.clone() if bag is None else bag * wrow
            tree_list, row_vals = grow_trees(codes, resid, bag, thresholds, cfg,
                                             return_row_values=True)
            tree = tree_list[0]
            trees.append(tree)
            weights.append(1.0 if rnd == 0 else step)
            # leaf-value gather (bit-identical to re-predicting the tree)
            margin = margin + weights[-1] * row_vals[0]
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

    @property
    def toDebugString(self) -> str:
        from flowml.classification.tree import ensemble_debug_string

        return ensemble_debug_string(
            f"GBTRegressionModel: uid={self.uid}, "
            f"numTrees={len(self._trees)}, numFeatures={self.numFeatures}",
            self._trees, self.treeWeights)

    @property
    def featureImportances(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._ensemble.feature_importances(self.numFeatures))

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        return self._ensemble.predict_raw(features)[:, 0]


# ---------------------------------------------------------------------------
# LinearRegression (spark:.../ml/regression/LinearRegression.scala)
# ---------------------------------------------------------------------------


from flowml.param import HasTol, HasWeightCol  # noqa: E402


class _LinearRegressionParams(_PredictorParams, HasMaxIter, HasTol, HasWeightCol):
    regParam: Param[float] = Param(
        "undefined", "regParam", "regularization parameter (>= 0).",
        TypeConverters.toFloat,
    )
    elasticNetParam: Param[float] = Param(
        "undefined", "elasticNetParam",
        "the ElasticNet mixing parameter, in range [0, 1] (0 = L2, 1 = L1).",
        TypeConverters.toFloat,
    )
    fitIntercept: Param[bool] = Param(
        "undefined", "fitIntercept", "whether to fit an intercept term.",
        TypeConverters.toBoolean,
    )
    standardization: Param[bool] = Param(
        "undefined", "standardization",
        "whether to standardize the training features before fitting.",
        TypeConverters.toBoolean,
    )


class LinearRegression(_LinearRegressionParams, Estimator, _RegressorIOMixin):
    """Least-squares linear regression with elastic-net regularization,
    solved by LBFGS / OWL-QN in standardized-feature space (the same
    optimizer stack as LogisticRegression; gradients all-reduce across
    ranks per iteration — SURVEY.md §2.5 C1 pattern).

    Documented divergence: Spark additionally scales the objective by the
    label std (so regParam is relative to a unit-variance label); here the
    objective is ½·Σw(ŷ−y)²/ΣW — identical solutions at regParam=0, and
    regParam differs by a yStd factor otherwise.
    """

    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(maxIter=100, regParam=0.0, elasticNetParam=0.0,
                         tol=1e-6, fitIntercept=True, standardization=True)
        self._set(**kwargs)

    def _fit(self, dataset: FlowFrame) -> "LinearRegressionModel":
        from flowml.optim import lbfgs, owlqn

        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol()).to(torch.float32)
        dev = X.device
        N, F = X.shape
        w = None
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float32)
        wcol = w if w is not None else torch.ones(N, device=dev)

        stats = torch.zeros(2 + 3 * F, dtype=torch.float64, device=dev)
        w64 = wcol.to(torch.float64)
        stats[0] = w64.sum()
        stats[1] = (w64 * y.to(torch.float64)).sum()
        X64 = X.to(torch.float64)
        stats[2 : 2 + F] = (X64 * w64.unsqueeze(1)).sum(dim=0)
        stats[2 + F : 2 + 2 * F] = (X64**2 * w64.unsqueeze(1)).sum(dim=0)
        stats[2 + 2 * F] = N
        dist.all_reduce_(stats)
        sumW = float(stats[0])
        y_mean = float(stats[1]) / max(sumW, 1e-300)
        mean = (stats[2 : 2 + F] / max(sumW, 1e-300)).to(torch.float32)
        var = (stats[2 + F : 2 + 2 * F] / max(sumW, 1e-300)).to(torch.float32) - mean**2
        totalN = float(stats[2 + 2 * F])
        if totalN > 1:
            var = var * (totalN / (totalN - 1.0))
        std = var.clamp_min(0).sqrt()
        inv_std = torch.where(std > 0, 1.0 / std, torch.zeros_like(std))
        Xs = X * inv_std

        fit_intercept = self.getOrDefault(self.fitIntercept)
        reg = self.getOrDefault(self.regParam)
        alpha = self.getOrDefault(self.elasticNetParam)
        standardization = self.getOrDefault(self.standardization)
        l2_w = torch.ones(F, device=dev) if standardization else inv_std**2
        l1_w = torch.ones(F, device=dev) if standardization else inv_std
        l2_coef = reg * (1.0 - alpha)
        l1_coef = reg * alpha
        n_params = F + (1 if fit_intercept else 0)
        theta0 = torch.zeros(n_params, dtype=torch.float32, device=dev)
        if fit_intercept:
            theta0[F] = y_mean

        use_hip = X.is_cuda and ops.hip_available(X)

        def f_grad(theta: torch.Tensor):
            beta = theta[:F]
            b = theta[F] if fit_intercept else torch.zeros((), device=dev)
            r = Xs @ beta + b - y
            wr = wcol * r if w is not None else r
            loss_s = 0.5 * (wr * r).sum() if w is not None else 0.5 * (r * r).sum()
            if use_hip:
                gW, gB = ops.get_ext().wgrad_f32(Xs, wr.unsqueeze(1).contiguous())
                gW = gW[0]
                gB = gB[0:1]
            else:
                gW = wr @ Xs
                gB = wr.sum().reshape(1)
            buf = torch.cat([loss_s.reshape(1), gW.reshape(-1), gB])
            dist.all_reduce_(buf)
            loss = float(buf[0]) / sumW
            gW = buf[1 : 1 + F] / sumW
            gB = buf[1 + F :] / sumW
            if l2_coef > 0:
                loss += 0.5 * l2_coef * float((l2_w * beta**2).sum())
                gW = gW + l2_coef * l2_w * beta
            grad = torch.cat([gW, gB]) if fit_intercept else gW
            return loss, grad

        max_iter = self.getMaxIter()
        tol = self.getTol()
        if l1_coef > 0:
            l1_vec = torch.zeros(n_params, device=dev)
            l1_vec[:F] = l1_coef * l1_w
            theta, history = owlqn(f_grad, theta0, l1_vec, max_iter=max_iter, tol=tol)
        else:
            theta, history = lbfgs(f_grad, theta0, max_iter=max_iter, tol=tol)

        coef = (theta[:F] * inv_std).detach().cpu()
        intercept = float(theta[F]) if fit_intercept else 0.0
        model = LinearRegressionModel(coefficients=coef, intercept=intercept,
                                      numFeatures=F)
        self._copyValues(model)
        model._set_parent(self)
        model._summary = LinearRegressionTrainingSummary(
            objectiveHistory=[float(h) for h in history],
            totalIterations=max(len(history) - 1, 0),
            model=model, dataset=dataset,
            labelCol=self.getLabelCol(),
            predictionCol=self.getOrDefault(self.predictionCol),
        )
        if reg == 0:
            # inference statistics (Spark exposes stdErrors/tValues/pValues
            # only for unregularized fits): OLS covariance from the
            # all-reduced normal equations on the RAW features
            Xd = X.to(torch.float64)
            P = F + (1 if fit_intercept else 0)
            Xa = torch.cat([Xd, torch.ones(N, 1, dtype=torch.float64,
                                           device=dev)], dim=1) if fit_intercept else Xd
            beta_raw = torch.cat([
                coef.to(torch.float64).to(dev),
                torch.tensor([intercept], dtype=torch.float64, device=dev)
            ]) if fit_intercept else coef.to(torch.float64).to(dev)
            r_vec = y.to(torch.float64) - Xa @ beta_raw
            wv = w.to(torch.float64) if w is not None else None
            Xw = Xa * wv.unsqueeze(1) if wv is not None else Xa
            rss = (wv * r_vec * r_vec).sum() if wv is not None else (r_vec * r_vec).sum()
            buf = torch.cat([(Xw.T @ Xa).reshape(-1), rss.reshape(1),
                             torch.tensor([float(N)], dtype=torch.float64,
                                          device=dev)])
            dist.all_reduce_(buf)
            try:
                import numpy as _np
                import scipy.stats as _st

                A = buf[: P * P].view(P, P)
                df_resid = max(float(buf[-1]) - P, 1.0)
                sigma2 = float(buf[P * P]) / df_resid
                cov = sigma2 * torch.linalg.inv(A)
                se = torch.sqrt(torch.diagonal(cov).clamp_min(0)).cpu().numpy()
                tv = beta_raw.cpu().numpy() / _np.maximum(se, 1e-300)
                pv = 2.0 * _st.t.sf(_np.abs(tv), df_resid)
                model._summary._inference = (
                    [float(v) for v in se], [float(v) for v in tv],
                    [float(v) for v in pv], int(df_resid))
            except Exception:
                pass
        return model


class LinearRegressionModel(_LinearRegressionParams, RegressionModel, _RegressorIOMixin):
    def __init__(self, coefficients=None, intercept: float = 0.0, numFeatures: int = 0):
        super().__init__()
        if coefficients is None:
            self._coef = torch.zeros(0)
        elif isinstance(coefficients, torch.Tensor):
            self._coef = coefficients.to(torch.float32)
        else:
            self._coef = torch.as_tensor(np_asarray_f32(coefficients))
        self.intercept = float(intercept)
        self.numFeatures = numFeatures
        self.scale = 1.0  # Spark field (non-huber -> 1.0)
        self._summary = None

    @property
    def coefficients(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._coef.to(torch.float64).numpy())

    def hasSummary(self) -> bool:
        return self._summary is not None

    @property
    def summary(self) -> "LinearRegressionTrainingSummary":
        if self._summary is None:
            raise RuntimeError("No training summary available for this model")
        return self._summary

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        c = self._coef.to(device=features.device, dtype=torch.float32)
        return features.to(torch.float32) @ c + self.intercept


class LinearRegressionTrainingSummary:
    """objectiveHistory + training-set regression metrics (Spark parity:
    rootMeanSquaredError, meanSquaredError, meanAbsoluteError, r2)."""

    def __init__(self, objectiveHistory, totalIterations, model=None,
                 dataset=None, labelCol="label", predictionCol="prediction"):
        self.objectiveHistory = objectiveHistory
        self.totalIterations = totalIterations
        self._model = model
        self._dataset = dataset
        self.labelCol = labelCol
        self.predictionCol = predictionCol
        self._pred = None
        self._inference = None  # (se, t, p, df) when regParam == 0

    def _infer_stats(self, i):
        if self._inference is None:
            raise RuntimeError(
                "No Std. Error of coefficients available for this "
                "LinearRegressionModel (regularized fit — Spark parity)")
        return self._inference[i]

    @property
    def coefficientStandardErrors(self):
        """OLS standard errors, coefficients first, intercept last
        (unregularized fits only — like Spark)."""
        return self._infer_stats(0)

    @property
    def tValues(self):
        return self._infer_stats(1)

    @property
    def pValues(self):
        return self._infer_stats(2)

    @property
    def degreesOfFreedom(self):
        return self._infer_stats(3)

    @property
    def predictions(self):
        if self._pred is None:
            self._pred = self._model.transform(self._dataset)
        return self._pred

    def _eval(self, name):
        from flowml.evaluation import RegressionEvaluator

        return RegressionEvaluator(metricName=name, labelCol=self.labelCol,
                                   predictionCol=self.predictionCol).evaluate(self.predictions)

    @property
    def rootMeanSquaredError(self):
        return self._eval("rmse")

    @property
    def meanSquaredError(self):
        return self._eval("mse")

    @property
    def meanAbsoluteError(self):
        return self._eval("mae")

    @property
    def r2(self):
        return self._eval("r2")


# ---------------------------------------------------------------------------
# IsotonicRegression (spark:.../ml/regression/IsotonicRegression.scala)
# ---------------------------------------------------------------------------


class IsotonicRegression(_PredictorParams, HasWeightCol, Estimator, _RegressorIOMixin):
    """Weighted isotonic (or antitonic) regression by pool-adjacent-violators;
    prediction linearly interpolates between fitted boundaries (Spark
    semantics, incl. the featureIndex selector for vector features)."""

    isotonic: Param[bool] = Param(
        "undefined", "isotonic",
        "whether the output sequence should be isotonic/increasing (true) "
        "or antitonic/decreasing (false)", TypeConverters.toBoolean,
    )
    featureIndex: Param[int] = Param(
        "undefined", "featureIndex",
        "the index of the feature if featuresCol is a vector column",
        TypeConverters.toInt,
    )

    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(isotonic=True, featureIndex=0)
        self._set(**kwargs)

    def _x_col(self, dataset: FlowFrame) -> torch.Tensor:
        x = dataset.col(self.getFeaturesCol())
        if isinstance(x, torch.Tensor) and x.dim() == 2:
            x = x[:, self.getOrDefault(self.featureIndex)]
        return x.reshape(-1).to(torch.float64)

    def _fit(self, dataset: FlowFrame) -> "IsotonicRegressionModel":
        import numpy as np

        x = self._x_col(dataset).cpu().numpy()
        y = _label_tensor(dataset, self.getLabelCol()).to(torch.float64).cpu().numpy()
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64).cpu().numpy()
        else:
            w = np.ones_like(y)
        inc = self.getOrDefault(self.isotonic)
        order = np.argsort(x, kind="stable")
        xs, ys, ws = x[order], y[order], w[order]
        if not inc:
            ys = -ys
        # PAVA with a block stack: (sum_wy, sum_w, x_lo, x_hi)
        blocks: list = []
        for xi, yi, wi in zip(xs, ys, ws):
            blocks.append([yi * wi, wi, xi, xi])
            while len(blocks) > 1 and (
                blocks[-2][0] / blocks[-2][1] >= blocks[-1][0] / blocks[-1][1]
            ):
                b = blocks.pop()
                blocks[-1][0] += b[0]
                blocks[-1][1] += b[1]
                blocks[-1][3] = b[3]
            # merge equal-x runs so boundaries stay strictly increasing
            while len(blocks) > 1 and blocks[-2][3] >= blocks[-1][2]:
                b = blocks.pop()
                blocks[-1][0] += b[0]
                blocks[-1][1] += b[1]
                blocks[-1][3] = b[3]
        bounds, preds = [], []
        for swy, sw, lo, hi in blocks:
            v = swy / sw
            bounds.append(lo)
            preds.append(v)
            if hi != lo:
                bounds.append(hi)
                preds.append(v)
        if not inc:
            preds = [-v for v in preds]
        model = IsotonicRegressionModel(
            boundaries=torch.tensor(bounds, dtype=torch.float64),
            predictions=torch.tensor(preds, dtype=torch.float64),
        )
        self._copyValues(model)
        model._set_parent(self)
        return model


class IsotonicRegressionModel(HasWeightCol, RegressionModel, _RegressorIOMixin):
    isotonic = IsotonicRegression.isotonic
    featureIndex = IsotonicRegression.featureIndex

    def __init__(self, boundaries: Optional[torch.Tensor] = None,
                 predictions: Optional[torch.Tensor] = None):
        super().__init__()
        self._bounds = boundaries if boundaries is not None else torch.zeros(0, dtype=torch.float64)
        self._preds = predictions if predictions is not None else torch.zeros(0, dtype=torch.float64)

    @property
    def boundaries(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._bounds.numpy())

    @property
    def predictions(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._preds.numpy())

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        x = features
        if x.dim() == 2:
            x = x[:, self.getOrDefault(self.featureIndex)]
        x = x.reshape(-1).to(torch.float64)
        b = self._bounds.to(x.device)
        p = self._preds.to(x.device)
        if len(b) == 0:
            return torch.zeros_like(x)
        if len(b) == 1:
            return p[0].expand_as(x).clone()
        idx = torch.searchsorted(b, x).clamp(1, len(b) - 1)
        lo, hi = b[idx - 1], b[idx]
        plo, phi = p[idx - 1], p[idx]
        t = torch.where(hi > lo, (x - lo) / (hi - lo), torch.zeros_like(x))
        out = plo + t.clamp(0, 1) * (phi - plo)
        out = torch.where(x <= b[0], p[0], out)
        out = torch.where(x >= b[-1], p[-1], out)
        return out

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getFeaturesCol())
        pred = self.predictRaw(x if isinstance(x, torch.Tensor) else torch.as_tensor(x))
        return dataset.withColumn(self.getOrDefault(self.predictionCol), pred)


# ---------------------------------------------------------------------------
# GeneralizedLinearRegression
# (spark:.../ml/regression/GeneralizedLinearRegression.scala)
# ---------------------------------------------------------------------------


_GLR_FAMILIES = {
    # family -> (default link, variance fn V(mu), mu init)
    "gaussian": ("identity", lambda mu: torch.ones_like(mu), lambda y: y),
    "binomial": ("logit", lambda mu: mu * (1 - mu),
                 lambda y: (y + 0.5) / 2.0),
    "poisson": ("log", lambda mu: mu, lambda y: y + 0.1),
    "gamma": ("inverse", lambda mu: mu * mu, lambda y: y),
}

_GLR_LINKS = {
    # link -> (g(mu), g^-1(eta), g'(mu))
    "identity": (lambda mu: mu, lambda e: e, lambda mu: torch.ones_like(mu)),
    "log": (lambda mu: torch.log(mu), lambda e: torch.exp(e),
            lambda mu: 1.0 / mu),
    "logit": (lambda mu: torch.log(mu / (1 - mu)), lambda e: torch.sigmoid(e),
              lambda mu: 1.0 / (mu * (1 - mu))),
    "inverse": (lambda mu: 1.0 / mu, lambda e: 1.0 / e,
                lambda mu: -1.0 / (mu * mu)),
    "sqrt": (lambda mu: torch.sqrt(mu), lambda e: e * e,
             lambda mu: 0.5 / torch.sqrt(mu)),
}


class _GLRParams(_PredictorParams, HasMaxIter, HasTol, HasWeightCol):
    regParam: Param[float] = Param(
        "undefined", "regParam", "regularization parameter (>= 0).",
        TypeConverters.toFloat)
    fitIntercept: Param[bool] = Param(
        "undefined", "fitIntercept", "whether to fit an intercept term.",
        TypeConverters.toBoolean)
    family: Param[str] = Param(
        "undefined", "family",
        "The name of family which is a description of the error distribution "
        "to be used in the model. Supported options: gaussian (default), "
        "binomial, poisson, gamma.", TypeConverters.toString)
    link: Param[str] = Param(
        "undefined", "link",
        "The name of link function which provides the relationship between "
        "the linear predictor and the mean of the distribution function. "
        "Supported options: identity, log, logit, inverse, sqrt.",
        TypeConverters.toString)
    linkPredictionCol: Param[str] = Param(
        "undefined", "linkPredictionCol",
        "link prediction (linear predictor) column name",
        TypeConverters.toString)

    def __init__(self):
        super().__init__()
        self._setDefault(family="gaussian", maxIter=25, tol=1e-6,
                         regParam=0.0, fitIntercept=True)


class GeneralizedLinearRegression(_GLRParams, Estimator, _RegressorIOMixin):
    """GLM fit by IRLS (spark:.../ml/regression/
    GeneralizedLinearRegression.scala): families gaussian/binomial/poisson/
    gamma with the standard links, maxIter=25, tol=1e-6 on the relative
    deviance change, regParam = L2 ridge on the IRLS normal equations.

    MI355X-first: each IRLS iteration accumulates the (F+1)² weighted
    normal-equation blocks XᵀWX / XᵀWz with two GEMMs over the resident
    shard, all-reduces the single flat buffer (F=41 ⇒ ~7 KB — SURVEY.md
    §2.5 C1 class), and every rank solves the tiny system identically —
    no per-iteration broadcast, same replicated-state pattern as
    LogisticRegression. (Tweedie is not implemented — documented.)"""

    def __init__(self, **kwargs):
        super().__init__()
        self._set(**kwargs)

    def setFamily(self, v: str):
        return self.set(self.family, v)

    def _fit(self, dataset: FlowFrame) -> "GeneralizedLinearRegressionModel":
        fam = self.getOrDefault(self.family)
        if fam not in _GLR_FAMILIES:
            raise ValueError(f"Unsupported family {fam!r} (tweedie not implemented)")
        default_link, varfn, mu0fn = _GLR_FAMILIES[fam]
        link = self.getOrDefault(self.link) if self.isSet(self.link) else default_link
        if link not in _GLR_LINKS:
            raise ValueError(f"Unsupported link {link!r}")
        g, ginv, gprime = _GLR_LINKS[link]
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float64)
        y = _label_tensor(dataset, self.getLabelCol()).to(torch.float64)
        N, F = X.shape
        w = None
        if self.isSet(self.weightCol):
            w = dataset.col(self.getWeightCol()).reshape(-1).to(torch.float64)
        fit_b = self.getOrDefault(self.fitIntercept)
        Xa = torch.cat([X, torch.ones(N, 1, dtype=torch.float64,
                                      device=X.device)], dim=1) if fit_b else X
        P = Xa.shape[1]
        reg = self.getOrDefault(self.regParam)
        tol = self.getOrDefault(self.tol)
        mu = mu0fn(y).clamp(1e-10, None)
        if fam == "binomial":
            mu = mu.clamp(1e-10, 1 - 1e-10)
        eta = g(mu)
        beta = None
        dev_prev = None
        it = 0
        for it in range(self.getMaxIter()):
            gp = gprime(mu)
            irls_w = 1.0 / (varfn(mu) * gp * gp).clamp_min(1e-300)
            if w is not None:
                irls_w = irls_w * w
            z = eta + (y - mu) * gp
            Xw = Xa * irls_w.unsqueeze(1)
            buf = torch.cat([(Xw.T @ Xa).reshape(-1), Xw.T @ z])
            dist.all_reduce_(buf)
            A = buf[: P * P].view(P, P)
            b = buf[P * P :]
            if reg > 0:
                ridge = reg * torch.eye(P, dtype=torch.float64, device=X.device)
                if fit_b:
                    ridge[P - 1, P - 1] = 0.0  # intercept unpenalized
                A = A + ridge
            beta = torch.linalg.solve(A, b)
            eta = Xa @ beta
            mu = ginv(eta)
            if fam == "binomial":
                mu = mu.clamp(1e-10, 1 - 1e-10)
            elif fam in ("poisson", "gamma"):
                mu = mu.clamp_min(1e-10)
            # deviance for convergence (family-specific, all-reduced)
            if fam == "gaussian":
                d = (y - mu) ** 2
            elif fam == "binomial":
                d = 2 * (torch.xlogy(y, y / mu) +
                         torch.xlogy(1 - y, (1 - y) / (1 - mu)))
            elif fam == "poisson":
                d = 2 * (torch.xlogy(y, y / mu) - (y - mu))
            else:  # gamma
                d = -2 * (torch.log((y / mu).clamp_min(1e-300)) - (y - mu) / mu)
            if w is not None:
                d = d * w
            dev_t = d.sum().reshape(1)
            dist.all_reduce_(dev_t)
            dev = float(dev_t)
            if dev_prev is not None and abs(dev - dev_prev) < tol * (
                abs(dev_prev) + 0.1
            ):
                dev_prev = dev
                break
            dev_prev = dev
        coef = beta[:F] if fit_b else beta
        intercept = float(beta[F]) if fit_b else 0.0
        model = GeneralizedLinearRegressionModel(
            coefficients=coef.to(torch.float64).cpu(), intercept=intercept,
            family=fam, link=link, deviance=dev_prev or 0.0, numIter=it + 1)
        # training summary (GeneralizedLinearRegressionTrainingSummary):
        # Fisher information at convergence -> covariance of beta.
        # Pearson X² = Σ w (y-mu)²/V(mu), dispersion fixed at 1 for
        # binomial/poisson else X²/df; z-test p-values when dispersion is
        # fixed, t-test otherwise (Spark's convention).
        gp = gprime(mu)
        irls_w = 1.0 / (varfn(mu) * gp * gp).clamp_min(1e-300)
        pearson = ((y - mu) ** 2 / varfn(mu).clamp_min(1e-300))
        if w is not None:
            irls_w = irls_w * w
            pearson = pearson * w
        Xw = Xa * irls_w.unsqueeze(1)
        buf2 = torch.cat([(Xw.T @ Xa).reshape(-1), pearson.sum().reshape(1),
                          torch.tensor([float(N)], dtype=torch.float64,
                                       device=X.device)])
        dist.all_reduce_(buf2)
        A_final = buf2[: P * P].view(P, P)
        pearson_x2 = float(buf2[P * P])
        n_tot = float(buf2[P * P + 1])
        df_resid = max(n_tot - P, 1.0)
        fixed_disp = fam in ("binomial", "poisson")
        dispersion = 1.0 if fixed_disp else pearson_x2 / df_resid
        try:
            cov = dispersion * torch.linalg.inv(A_final)
            se = torch.sqrt(torch.diagonal(cov).clamp_min(0))
            tvals = beta / se.clamp_min(1e-300)
            import numpy as np
            import scipy.stats as _st

            tv = tvals.cpu().numpy()
            if fixed_disp:
                pv = 2.0 * _st.norm.sf(np.abs(tv))
            else:
                pv = 2.0 * _st.t.sf(np.abs(tv), df_resid)
            model._summary = GeneralizedLinearRegressionTrainingSummary(
                coefficientStandardErrors=[float(v) for v in se],
                tValues=[float(v) for v in tv],
                pValues=[float(v) for v in pv],
                dispersion=float(dispersion), deviance=float(dev_prev or 0.0),
                residualDegreeOfFreedom=int(df_resid),
                numIterations=it + 1)
        except Exception as _e:  # singular information matrix: no summary
            import os as _os
            if _os.environ.get("FLOWML_DEBUG", "0") == "1":
                raise
            model._summary = None
        self._copyValues(model)
        model._set_parent(self)
        return model


class GeneralizedLinearRegressionTrainingSummary:
    """Spark GeneralizedLinearRegressionTrainingSummary subset: inference
    statistics from the Fisher information at convergence. Ordering follows
    Spark: coefficients first, intercept LAST."""

    def __init__(self, coefficientStandardErrors, tValues, pValues,
                 dispersion, deviance, residualDegreeOfFreedom, numIterations):
        self.coefficientStandardErrors = coefficientStandardErrors
        self.tValues = tValues
        self.pValues = pValues
        self.dispersion = dispersion
        self.deviance = deviance
        self.residualDegreeOfFreedom = residualDegreeOfFreedom
        self.numIterations = numIterations


class GeneralizedLinearRegressionModel(_GLRParams, Model, _RegressorIOMixin):
    def __init__(self, coefficients: Optional[torch.Tensor] = None,
                 intercept: float = 0.0, family: str = "gaussian",
                 link: str = "identity", deviance: float = 0.0,
                 numIter: int = 0):
        super().__init__()
        self._coef = coefficients if coefficients is not None else torch.zeros(0, dtype=torch.float64)
        self.intercept = float(intercept)
        self._family = family
        self._link = link
        self.deviance = float(deviance)
        self.numIter = int(numIter)
        self._summary = None
        self._set(family=family, link=link)

    @property
    def summary(self) -> "GeneralizedLinearRegressionTrainingSummary":
        if self._summary is None:
            raise RuntimeError(
                "No training summary available (absent after load, like Spark)")
        return self._summary

    def hasSummary(self) -> bool:
        return self._summary is not None

    @property
    def coefficients(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._coef.numpy())

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getFeaturesCol()).to(torch.float64)
        eta = x @ self._coef.to(x.device) + self.intercept
        mu = _GLR_LINKS[self._link][1](eta)
        out = dataset.withColumn(self.getOrDefault(self.predictionCol),
                                 mu.to(torch.float64))
        if self.isSet(self.linkPredictionCol):
            out = out.withColumn(self.getOrDefault(self.linkPredictionCol),
                                 eta.to(torch.float64))
        return out


__all__ = [
    "RegressionModel",
    "GeneralizedLinearRegression", "GeneralizedLinearRegressionModel",
    "GeneralizedLinearRegressionTrainingSummary",
    "LinearRegression", "LinearRegressionModel", "LinearRegressionTrainingSummary",
    "DecisionTreeRegressor", "DecisionTreeRegressionModel",
    "RandomForestRegressor", "RandomForestRegressionModel",
    "GBTRegressor", "GBTRegressionModel",
    "IsotonicRegression", "IsotonicRegressionModel",
]
__all__ += ["FMRegressor", "FMRegressionModel"]


def __getattr__(name):  # PEP 562: lazy FM re-export (flowml.fm imports us)
    if name in ("FMRegressor", "FMRegressionModel"):
        import flowml.fm as _fm

        return getattr(_fm, name)
    raise AttributeError(f"module {__name__!r} has no attribute {name!r}")
