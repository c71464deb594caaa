"""Spark ML persistence: save/load in the Spark ML directory layout.

Format contract (BASELINE.json:5 "model-save format"; SURVEY.md §3.5 and
§2.2 persistence row; upstream ``spark:.../ml/util/ReadWrite.scala``):

    <path>/
      metadata/part-00000      one-line JSON {class, timestamp, sparkVersion,
                               uid, paramMap, defaultParamMap}
      data/part-*.parquet      learned state (model-specific schema below)
      stages/<idx>_<uid>/      (PipelineModel / Pipeline only) nested stage dirs
      treesMetadata/           (ensembles) per-tree {treeID, metadata, weights}

Model ``data/`` schemas:
  StringIndexerModel       {labelsArray: list<list<string>>}
  OneHotEncoderModel       {categorySizes: list<int>}
  StandardScalerModel      {std: VectorUDT, mean: VectorUDT}
  MinMaxScalerModel        {originalMin: VectorUDT, originalMax: VectorUDT}
  LogisticRegressionModel  {numClasses, numFeatures, interceptVector: VectorUDT,
                            coefficientMatrix: MatrixUDT, isMultinomial}
  MLPClassificationModel   {weights: VectorUDT}   (layers live in paramMap)
  DecisionTree/RF/GBT      NodeData rows {treeID, nodeData: {id, prediction,
                            impurity, impurityStats[], gain, leftChild,
                            rightChild, split: {featureIndex,
                            leftCategoriesOrThreshold[], numCategories}}}

VectorUDT / MatrixUDT are encoded as the Spark struct layouts
(type/size/indices/values and type/numRows/numCols/colPtrs/rowIndices/values/
isTransposed). ``class`` uses the org.apache.spark.ml.* names so the layout
matches what a Spark reader expects; ``sparkVersion`` is "3.5.0-flowml".
Fields whose exact upstream value could not be confirmed without a live Spark
(SURVEY.md §7 hard-part 2) are marked in comments.
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, List, Optional

import numpy as np
import torch

SPARK_VERSION = "3.5.0-flowml"


# ---------------------------------------------------------------------------
# class-name mapping
# ---------------------------------------------------------------------------

_CLASS_TO_SPARK: Dict[str, str] = {}
_SPARK_TO_CLASS: Dict[str, Any] = {}


def _register(cls_name: str, spark_name: str):
    _CLASS_TO_SPARK[cls_name] = spark_name


_register("Pipeline", "org.apache.spark.ml.Pipeline")
_register("PipelineModel", "org.apache.spark.ml.PipelineModel")
_register("StringIndexer", "org.apache.spark.ml.feature.StringIndexer")
_register("StringIndexerModel", "org.apache.spark.ml.feature.StringIndexerModel")
_register("OneHotEncoder", "org.apache.spark.ml.feature.OneHotEncoder")
_register("OneHotEncoderModel", "org.apache.spark.ml.feature.OneHotEncoderModel")
_register("VectorAssembler", "org.apache.spark.ml.feature.VectorAssembler")
_register("IndexToString", "org.apache.spark.ml.feature.IndexToString")
_register("Normalizer", "org.apache.spark.ml.feature.Normalizer")
_register("Binarizer", "org.apache.spark.ml.feature.Binarizer")
_register("StandardScaler", "org.apache.spark.ml.feature.StandardScaler")
_register("StandardScalerModel", "org.apache.spark.ml.feature.StandardScalerModel")
_register("MinMaxScaler", "org.apache.spark.ml.feature.MinMaxScaler")
_register("MaxAbsScaler", "org.apache.spark.ml.feature.MaxAbsScaler")
_register("MaxAbsScalerModel", "org.apache.spark.ml.feature.MaxAbsScalerModel")
_register("Bucketizer", "org.apache.spark.ml.feature.Bucketizer")
_register("QuantileDiscretizer", "org.apache.spark.ml.feature.QuantileDiscretizer")
_register("PCA", "org.apache.spark.ml.feature.PCA")
_register("PCAModel", "org.apache.spark.ml.feature.PCAModel")
_register("VectorSlicer", "org.apache.spark.ml.feature.VectorSlicer")
_register("VectorIndexer", "org.apache.spark.ml.feature.VectorIndexer")
_register("RFormula", "org.apache.spark.ml.feature.RFormula")
_register("VarianceThresholdSelector", "org.apache.spark.ml.feature.VarianceThresholdSelector")
_register("VarianceThresholdSelectorModel", "org.apache.spark.ml.feature.VarianceThresholdSelectorModel")
_register("VectorSizeHint", "org.apache.spark.ml.feature.VectorSizeHint")
_register("RFormulaModel", "org.apache.spark.ml.feature.RFormulaModel")
_register("RobustScaler", "org.apache.spark.ml.feature.RobustScaler")
_register("UnivariateFeatureSelector", "org.apache.spark.ml.feature.UnivariateFeatureSelector")
_register("UnivariateFeatureSelectorModel", "org.apache.spark.ml.feature.UnivariateFeatureSelectorModel")
_register("ChiSqSelector", "org.apache.spark.ml.feature.ChiSqSelector")
_register("RobustScalerModel", "org.apache.spark.ml.feature.RobustScalerModel")
_register("ElementwiseProduct", "org.apache.spark.ml.feature.ElementwiseProduct")
_register("PolynomialExpansion", "org.apache.spark.ml.feature.PolynomialExpansion")
_register("VectorIndexerModel", "org.apache.spark.ml.feature.VectorIndexerModel")
_register("Interaction", "org.apache.spark.ml.feature.Interaction")
_register("KMeans", "org.apache.spark.ml.clustering.KMeans")
_register("KMeansModel", "org.apache.spark.ml.clustering.KMeansModel")
_register("BisectingKMeans", "org.apache.spark.ml.clustering.BisectingKMeans")
_register("BisectingKMeansModel", "org.apache.spark.ml.clustering.BisectingKMeansModel")
_register("GaussianMixture", "org.apache.spark.ml.clustering.GaussianMixture")
_register("GaussianMixtureModel", "org.apache.spark.ml.clustering.GaussianMixtureModel")
_register("CrossValidatorModel", "org.apache.spark.ml.tuning.CrossValidatorModel")
_register("TrainValidationSplitModel", "org.apache.spark.ml.tuning.TrainValidationSplitModel")
_register("OneVsRestModel", "org.apache.spark.ml.classification.OneVsRestModel")
_register("FMClassifier", "org.apache.spark.ml.classification.FMClassifier")
_register("FMClassificationModel", "org.apache.spark.ml.classification.FMClassificationModel")
_register("FMRegressor", "org.apache.spark.ml.regression.FMRegressor")
_register("FMRegressionModel", "org.apache.spark.ml.regression.FMRegressionModel")
_register("IsotonicRegression", "org.apache.spark.ml.regression.IsotonicRegression")
_register("IsotonicRegressionModel", "org.apache.spark.ml.regression.IsotonicRegressionModel")
_register("Imputer", "org.apache.spark.ml.feature.Imputer")
_register("RandomForestRegressor", "org.apache.spark.ml.regression.RandomForestRegressor")
_register("RandomForestRegressionModel", "org.apache.spark.ml.regression.RandomForestRegressionModel")
_register("DecisionTreeRegressor", "org.apache.spark.ml.regression.DecisionTreeRegressor")
_register("DecisionTreeRegressionModel", "org.apache.spark.ml.regression.DecisionTreeRegressionModel")
_register("GBTRegressor", "org.apache.spark.ml.regression.GBTRegressor")
_register("GBTRegressionModel", "org.apache.spark.ml.regression.GBTRegressionModel")
_register("LinearRegression", "org.apache.spark.ml.regression.LinearRegression")
_register("GeneralizedLinearRegression", "org.apache.spark.ml.regression.GeneralizedLinearRegression")
_register("GeneralizedLinearRegressionModel", "org.apache.spark.ml.regression.GeneralizedLinearRegressionModel")
_register("LinearRegressionModel", "org.apache.spark.ml.regression.LinearRegressionModel")
_register("NaiveBayes", "org.apache.spark.ml.classification.NaiveBayes")
_register("NaiveBayesModel", "org.apache.spark.ml.classification.NaiveBayesModel")
_register("LinearSVC", "org.apache.spark.ml.classification.LinearSVC")
_register("LinearSVCModel", "org.apache.spark.ml.classification.LinearSVCModel")
_register("ImputerModel", "org.apache.spark.ml.feature.ImputerModel")
_register("MinMaxScalerModel", "org.apache.spark.ml.feature.MinMaxScalerModel")
_register("LogisticRegression", "org.apache.spark.ml.classification.LogisticRegression")
_register("LogisticRegressionModel", "org.apache.spark.ml.classification.LogisticRegressionModel")
_register("RandomForestClassifier", "org.apache.spark.ml.classification.RandomForestClassifier")
_register(
    "RandomForestClassificationModel",
    "org.apache.spark.ml.classification.RandomForestClassificationModel",
)
_register("GBTClassifier", "org.apache.spark.ml.classification.GBTClassifier")
_register("GBTClassificationModel", "org.apache.spark.ml.classification.GBTClassificationModel")
_register(
    "DecisionTreeClassificationModel",
    "org.apache.spark.ml.classification.DecisionTreeClassificationModel",
)
_register(
    "MultilayerPerceptronClassifier",
    "org.apache.spark.ml.classification.MultilayerPerceptronClassifier",
)
_register(
    "MultilayerPerceptronClassificationModel",
    "org.apache.spark.ml.classification.MultilayerPerceptronClassificationModel",
)


def _our_class(spark_name: str):
    import flowml.feature as feature
    import flowml.pipeline as pipeline
    import flowml.classification as cls
    import flowml.regression as reg
    import flowml.clustering as clus
    import flowml.tuning as tuning
    import flowml.fm as fm

    simple = spark_name.rsplit(".", 1)[-1]
    for mod in (pipeline, feature, cls, reg, clus, tuning, fm):
        if hasattr(mod, simple):
            return getattr(mod, simple)
    raise ValueError(f"Cannot resolve class for {spark_name!r}")


# ---------------------------------------------------------------------------
# metadata read/write
# ---------------------------------------------------------------------------


def _json_value(v: Any) -> Any:
    if isinstance(v, (np.integer,)):
        return int(v)
    if isinstance(v, (np.floating,)):
        return float(v)
    if isinstance(v, (list, tuple)):
        return [_json_value(x) for x in v]
    return v


def save_metadata(instance, path: str, extra: Optional[Dict[str, Any]] = None) -> None:
    os.makedirs(os.path.join(path, "metadata"), exist_ok=True)
    cls_simple = type(instance).__name__
    spark_class = _CLASS_TO_SPARK.get(cls_simple, f"flowml.{cls_simple}")
    paramMap = {}
    for p, v in instance._paramMap.items():
        if p.name == "stages":
            continue  # Pipeline stages serialize structurally
        paramMap[p.name] = _json_value(v)
    defaultParamMap = {
        p.name: _json_value(v) for p, v in instance._defaultParamMap.items()
    }
    meta = {
        "class": spark_class,
        "timestamp": int(time.time() * 1000),
        "sparkVersion": SPARK_VERSION,
        "uid": instance.uid,
        "paramMap": paramMap,
        "defaultParamMap": defaultParamMap,
    }
    if extra:
        meta.update(extra)
    with open(os.path.join(path, "metadata", "part-00000"), "w") as f:
        f.write(json.dumps(meta) + "\n")
    open(os.path.join(path, "metadata", "_SUCCESS"), "w").close()


def read_metadata(path: str) -> Dict[str, Any]:
    with open(os.path.join(path, "metadata", "part-00000")) as f:
        return json.loads(f.readline())


def _apply_params(instance, meta: Dict[str, Any]) -> None:
    instance.uid = meta["uid"]
    # re-parent instance params so set/get keep working after the uid change
    from flowml.param import Param

    for name, value in list(instance.__dict__.items()):
        if isinstance(value, Param):
            value.parent = instance.uid
    # re-parenting changed the Params' hashes — rebuild the maps so dict
    # lookups work again
    instance._paramMap = {p: v for p, v in instance._paramMap.items()}
    instance._defaultParamMap = {p: v for p, v in instance._defaultParamMap.items()}
    for name, v in meta.get("defaultParamMap", {}).items():
        if instance.hasParam(name):
            p = instance.getParam(name)
            if v is not None and p.typeConverter is not None:
                v = p.typeConverter(v)
            instance._defaultParamMap[p] = v
    for name, v in meta.get("paramMap", {}).items():
        if instance.hasParam(name):
            p = instance.getParam(name)
            if v is not None and p.typeConverter is not None:
                v = p.typeConverter(v)
            instance._paramMap[p] = v


# ---------------------------------------------------------------------------
# parquet helpers (VectorUDT / MatrixUDT structs)
# ---------------------------------------------------------------------------


def _write_parquet(path: str, table) -> None:
    import pyarrow.parquet as pq

    os.makedirs(path, exist_ok=True)
    pq.write_table(table, os.path.join(path, "part-00000.parquet"))
    open(os.path.join(path, "_SUCCESS"), "w").close()


def _read_parquet(path: str):
    import pyarrow.parquet as pq

    parts = sorted(
        f for f in os.listdir(path) if f.endswith(".parquet") or f.startswith("part-")
    )
    parts = [f for f in parts if not f.startswith("_")]
    import pyarrow as pa

    tables = [pq.read_table(os.path.join(path, f)) for f in parts]
    return pa.concat_tables(tables)


def vector_struct(values: np.ndarray) -> Dict[str, Any]:
    """Dense VectorUDT struct row (type=1)."""
    return {
        "type": 1,
        "size": None,
        "indices": None,
        "values": [float(v) for v in np.asarray(values, dtype=np.float64)],
    }


def vector_from_struct(s: Dict[str, Any]) -> np.ndarray:
    if s["type"] == 1:
        return np.asarray(s["values"], dtype=np.float64)
    size = int(s["size"])
    arr = np.zeros(size, dtype=np.float64)
    idx = np.asarray(s["indices"], dtype=np.int64)
    arr[idx] = np.asarray(s["values"], dtype=np.float64)
    return arr


def matrix_struct(m: np.ndarray) -> Dict[str, Any]:
    """Dense MatrixUDT struct row (type=1, column-major values)."""
    m = np.asarray(m, dtype=np.float64)
    return {
        "type": 1,
        "numRows": int(m.shape[0]),
        "numCols": int(m.shape[1]),
        "colPtrs": None,
        "rowIndices": None,
        "values": [float(v) for v in m.T.ravel()],
        "isTransposed": False,
    }


def matrix_from_struct(s: Dict[str, Any]) -> np.ndarray:
    vals = np.asarray(s["values"], dtype=np.float64)
    r, c = int(s["numRows"]), int(s["numCols"])
    if s.get("isTransposed"):
        return vals.reshape(r, c)
    return vals.reshape(c, r).T


def _table_from_rows(rows: List[Dict[str, Any]]):
    import pyarrow as pa

    return pa.Table.from_pylist(rows)


# ---------------------------------------------------------------------------
# per-class writers
# ---------------------------------------------------------------------------


def _save_instance(instance, path: str) -> None:
    cls = type(instance).__name__
    os.makedirs(path, exist_ok=True)
    if cls == "RFormulaModel":
        # Spark nests the fitted featurization inside the model dir; here:
        # metadata (params + resolved terms) + stages/<i>_<uid>/ + labelStage/
        save_metadata(instance, path, extra={
            "resolvedTerms": [list(t) if isinstance(t, tuple) else t
                              for t in instance._terms],
            "assembleCols": list(instance._assemble),
            "labelSource": instance._label_source,
            "hasLabelStage": instance._label_stage is not None,
        })
        _save_pipeline_stages(instance._stages, path)
        if instance._label_stage is not None:
            _save_instance(instance._label_stage,
                           os.path.join(path, "labelStage"))
        return
    if cls == "Pipeline":
        _save_pipeline_stages(instance.getStages(), path)
        save_metadata(instance, path)
        return
    if cls == "PipelineModel":
        _save_pipeline_stages(instance.stages, path)
        save_metadata(instance, path)
        return
    if cls == "OneVsRestModel":
        # Spark layout: metadata/ + models/<i>_<uid>/ per binary sub-model
        save_metadata(instance, path)
        models_dir = os.path.join(path, "models")
        os.makedirs(models_dir, exist_ok=True)
        for i, m in enumerate(instance.models):
            _save_instance(m, os.path.join(models_dir, f"{i}_{m.uid}"))
        return
    if cls in ("CrossValidatorModel", "TrainValidationSplitModel"):
        # Spark layout: metadata/ + bestModel/ subdir (we persist the
        # metrics in data/; the estimator/evaluator/param-maps needed only
        # for refitting are not persisted — documented divergence)
        save_metadata(instance, path)
        _save_instance(instance.bestModel, os.path.join(path, "bestModel"))
        metrics = (instance.avgMetrics if cls == "CrossValidatorModel"
                   else instance.validationMetrics)
        _write_parquet(os.path.join(path, "data"),
                       _table_from_rows([{"metrics": [float(m) for m in metrics]}]))
        return
    save_metadata(instance, path)
    rows = _data_rows(instance, path)
    if rows is not None:
        _write_parquet(os.path.join(path, "data"), _table_from_rows(rows))
    else:
        from flowml.base import Model

        if isinstance(instance, Model):
            # every fitted Model must have an explicit data writer —
            # silently persisting only the params would lose learned state
            raise NotImplementedError(
                f"No persistence writer for fitted model {cls}; "
                "refusing to save params-only"
            )


def _save_pipeline_stages(stages, path: str) -> None:
    stages_dir = os.path.join(path, "stages")
    os.makedirs(stages_dir, exist_ok=True)
    for i, stage in enumerate(stages):
        stage_dir = os.path.join(stages_dir, f"{i}_{stage.uid}")
        _save_instance(stage, stage_dir)


def _tree_node_rows(tree, tree_id: int) -> List[Dict[str, Any]]:
    """Pre-order NodeData rows (SURVEY.md §2.2 persistence schema)."""
    rows: List[Dict[str, Any]] = []
    order: List[int] = []
    new_id: Dict[int, int] = {}

    def visit(i: int):
        new_id[i] = len(order)
        order.append(i)
        if tree.feature[i] >= 0:
            visit(tree.left[i])
            visit(tree.right[i])

    if tree.n_nodes:
        visit(0)
    for i in order:
        internal = tree.feature[i] >= 0
        mask = int(tree.cat_mask[i]) if internal else 0
        if mask:
            # categorical split: the left category set + the feature arity
            # (Spark CategoricalSplit serialization)
            lcot = [float(c) for c in range(64) if (mask >> c) & 1]
            ncat = int(tree.num_cats[i])
        elif internal:
            lcot = [float(tree.thr_raw[i])]  # continuous: single threshold
            ncat = 0
        else:
            lcot, ncat = [], -1
        rows.append(
            {
                "treeID": tree_id,
                "nodeData": {
                    "id": new_id[i],
                    "prediction": float(tree.prediction(i)),
                    "impurity": float(tree.impurity[i]),
                    "impurityStats": [float(v) for v in tree.stats[i]],
                    "gain": float(tree.gain[i]) if internal else -1.0,
                    "leftChild": new_id[tree.left[i]] if internal else -1,
                    "rightChild": new_id[tree.right[i]] if internal else -1,
                    "split": {
                        "featureIndex": int(tree.feature[i]) if internal else -1,
                        "leftCategoriesOrThreshold": lcot,
                        "numCategories": ncat,
                    },
                },
            }
        )
    return rows


def _trees_from_node_rows(rows: List[Dict[str, Any]], regression: bool = False):
    """Rebuild FlatTree list from NodeData rows.

    ``regression`` is threaded from the model class being loaded (regression
    trees — incl. GBT classification's internal trees — store (s, s2, n)
    stats; classification stores per-class counts). Width-3 stats are
    ambiguous between the two, so this must never be shape-sniffed.
    """
    from flowml.classification.tree import FlatTree

    by_tree: Dict[int, List[Dict[str, Any]]] = {}
    for r in rows:
        by_tree.setdefault(int(r["treeID"]), []).append(r["nodeData"])
    trees = []
    for tid in sorted(by_tree):
        nodes = sorted(by_tree[tid], key=lambda nd: nd["id"])
        tr = FlatTree(regression=regression)
        for nd in nodes:
            stats = np.asarray(nd["impurityStats"], dtype=np.float64)
            cnt = float(stats[2] if regression else stats.sum())
            tr.add_node(stats, float(nd["impurity"]), cnt)
        for nd in nodes:
            i = nd["id"]
            if nd["leftChild"] >= 0:
                sp = nd["split"]
                tr.feature[i] = int(sp["featureIndex"])
                ncat = int(sp.get("numCategories", 0))
                if ncat > 0:
                    mask = 0
                    for c in sp["leftCategoriesOrThreshold"]:
                        mask |= 1 << int(c)
                    tr.cat_mask[i] = mask
                    tr.num_cats[i] = ncat
                    tr.thr_bin[i] = -1
                    tr.thr_raw[i] = float("nan")
                else:
                    tr.thr_raw[i] = float(sp["leftCategoriesOrThreshold"][0])
                tr.left[i] = int(nd["leftChild"])
                tr.right[i] = int(nd["rightChild"])
                tr.gain[i] = float(nd["gain"])
        trees.append(tr)
    return trees


def _data_rows(instance, path: str) -> Optional[List[Dict[str, Any]]]:
    cls = type(instance).__name__
    if cls == "StringIndexerModel":
        return [{"labelsArray": [list(l) for l in instance.labelsArray]}]
    if cls == "OneHotEncoderModel":
        return [{"categorySizes": [int(s) for s in instance.categorySizes]}]
    if cls == "StandardScalerModel":
        return [
            {
                "std": vector_struct(instance.std.numpy()),
                "mean": vector_struct(instance.mean.numpy()),
            }
        ]
    if cls == "MinMaxScalerModel":
        return [
            {
                "originalMin": vector_struct(instance.originalMin.numpy()),
                "originalMax": vector_struct(instance.originalMax.numpy()),
            }
        ]
    if cls == "VarianceThresholdSelectorModel":
        return [{"selectedFeatures": [int(i) for i in instance.selectedFeatures]}]
    if cls == "UnivariateFeatureSelectorModel":
        return [{"selectedFeatures": [int(i) for i in instance.selectedFeatures]}]
    if cls == "RobustScalerModel":
        return [
            {
                "range": vector_struct(instance.range.numpy()),
                "median": vector_struct(instance.median.numpy()),
            }
        ]
    if cls == "MaxAbsScalerModel":
        return [{"maxAbs": vector_struct(instance.maxAbs.numpy())}]
    if cls == "VectorIndexerModel":
        # Spark stores Data(numFeatures, categoryMaps: Map[Int, Map[Double,
        # Int]]); parquet map-of-map types are fragile across readers, so
        # categoryMaps is flattened to one row per categorical feature with
        # the category values ordered by their index (index = position) —
        # marked as an unconfirmed-field divergence (SURVEY.md §7 part 2).
        rows = [{"numFeatures": int(instance.numFeatures),
                 "featureIndex": -1, "categories": []}]
        for j in sorted(instance.categoryMaps):
            cmap = instance.categoryMaps[j]
            ordered = [v for v, _ in sorted(cmap.items(), key=lambda t: t[1])]
            rows.append({"numFeatures": int(instance.numFeatures),
                         "featureIndex": int(j),
                         "categories": [float(v) for v in ordered]})
        return rows
    if cls == "ImputerModel":
        return [{"surrogates": [float(v) for v in instance.surrogates]}]
    if cls == "LogisticRegressionModel":
        return [
            {
                "numClasses": int(instance.numClasses),
                "numFeatures": int(instance.numFeatures),
                "interceptVector": vector_struct(instance._b.to(torch.float64).cpu().numpy()),
                "coefficientMatrix": matrix_struct(instance._W.to(torch.float64).cpu().numpy()),
                "isMultinomial": bool(instance.isMultinomial),
            }
        ]
    if cls == "NaiveBayesModel":
        return [
            {
                "pi": vector_struct(instance._pi.to(torch.float64).cpu().numpy()),
                "theta": matrix_struct(instance._theta.to(torch.float64).cpu().numpy()),
                "sigma": matrix_struct(
                    instance._sigma.to(torch.float64).cpu().numpy().reshape(
                        instance._theta.shape if instance._sigma.numel() else (0, 0))),
                "modelType": instance._mtype,
            }
        ]
    if cls == "BisectingKMeansModel":
        return [{"clusterCenters": matrix_struct(instance._centers.numpy()),
                 "trainingCost": float(instance.trainingCost)}]
    if cls == "GaussianMixtureModel":
        # Spark schema: single row {weights: array<double>,
        # mus: array<VectorUDT>, sigmas: array<MatrixUDT>}
        return [
            {
                "weights": [float(v) for v in instance._weights],
                "mus": [vector_struct(instance._mus[j].numpy())
                        for j in range(instance._mus.shape[0])],
                "sigmas": [matrix_struct(instance._covs[j].numpy())
                           for j in range(instance._covs.shape[0])],
            }
        ]
    if cls == "KMeansModel":
        return [
            {
                "clusterCenters": matrix_struct(instance._centers.numpy()),
                "trainingCost": float(instance.trainingCost),
            }
        ]
    if cls == "PCAModel":
        return [
            {
                "pc": matrix_struct(instance._pc.numpy()),
                "explainedVariance": vector_struct(instance._ev.numpy()),
            }
        ]
    if cls == "IsotonicRegressionModel":
        return [
            {
                "boundaries": vector_struct(instance._bounds.numpy()),
                "predictions": vector_struct(instance._preds.numpy()),
            }
        ]
    if cls in ("FMClassificationModel", "FMRegressionModel"):
        return [
            {
                "intercept": float(instance.intercept),
                "linear": vector_struct(instance._linear.to(torch.float64).cpu().numpy()),
                "factors": matrix_struct(instance._factors.to(torch.float64).cpu().numpy()),
            }
        ]
    if cls == "LinearSVCModel":
        return [
            {
                "intercept": float(instance.intercept),
                "coefficients": vector_struct(instance._coef.to(torch.float64).cpu().numpy()),
            }
        ]
    if cls == "GeneralizedLinearRegressionModel":
        return [
            {
                "intercept": float(instance.intercept),
                "coefficients": vector_struct(instance._coef.to(torch.float64).cpu().numpy()),
            }
        ]
    if cls == "LinearRegressionModel":
        return [
            {
                "intercept": float(instance.intercept),
                "coefficients": vector_struct(instance._coef.to(torch.float64).cpu().numpy()),
                "scale": float(instance.scale),
            }
        ]
    if cls == "MultilayerPerceptronClassificationModel":
        return [{"weights": vector_struct(instance._weights.to(torch.float64).cpu().numpy())}]
    if cls in ("RandomForestClassificationModel", "GBTClassificationModel",
               "RandomForestRegressionModel", "GBTRegressionModel",
               "DecisionTreeRegressionModel"):
        rows: List[Dict[str, Any]] = []
        for tid, tree in enumerate(instance._trees):
            rows.extend(_tree_node_rows(tree, tid))
        # treesMetadata: per-tree {treeID, metadata(json), weights}
        tw = (
            instance.treeWeights
            if cls in ("GBTClassificationModel", "GBTRegressionModel")
            else [1.0] * len(instance._trees)
        )
        tmeta = [
            {"treeID": t, "metadata": json.dumps({"numFeatures": instance.numFeatures}), "weights": float(w)}
            for t, w in enumerate(tw)
        ]
        _write_parquet(os.path.join(path, "treesMetadata"), _table_from_rows(tmeta))
        return rows
    if cls == "DecisionTreeClassificationModel":
        return _tree_node_rows(instance._tree, 0)
    # pure-param transformers/estimators (VectorAssembler, StringIndexer, ...)
    return None


# ---------------------------------------------------------------------------
# loaders
# ---------------------------------------------------------------------------


def load(path: str):
    meta = read_metadata(path)
    spark_class = meta["class"]
    cls = _our_class(spark_class)
    simple = cls.__name__

    if simple == "OneVsRestModel":
        models_dir = os.path.join(path, "models")
        entries = sorted(os.listdir(models_dir), key=lambda s: int(s.split("_", 1)[0]))
        models = [load(os.path.join(models_dir, e)) for e in entries]
        inst = cls(models=models, numClasses=len(models))
        _apply_params(inst, meta)
        return inst

    if simple in ("CrossValidatorModel", "TrainValidationSplitModel"):
        best = load(os.path.join(path, "bestModel"))
        rows = _read_parquet(os.path.join(path, "data")).to_pylist()
        metrics = [float(m) for m in rows[0]["metrics"]]
        if simple == "CrossValidatorModel":
            inst = cls(bestModel=best, avgMetrics=metrics)
        else:
            inst = cls(bestModel=best, validationMetrics=metrics)
        _apply_params(inst, meta)
        return inst

    if simple == "RFormulaModel":
        stages_dir = os.path.join(path, "stages")
        stages = []
        if os.path.isdir(stages_dir):
            entries = sorted(os.listdir(stages_dir),
                             key=lambda e: int(e.split("_", 1)[0]))
            stages = [load(os.path.join(stages_dir, e)) for e in entries]
        label_stage = None
        if meta.get("hasLabelStage") and os.path.isdir(
            os.path.join(path, "labelStage")
        ):
            label_stage = load(os.path.join(path, "labelStage"))
        terms = [tuple(t) if isinstance(t, list) else t
                 for t in meta.get("resolvedTerms", [])]
        inst = cls(terms=terms, stages=stages,
                   assemble_cols=list(meta.get("assembleCols", [])),
                   label_source=meta.get("labelSource", ""),
                   label_stage=label_stage)
        _apply_params(inst, meta)
        return inst

    if simple in ("Pipeline", "PipelineModel"):
        stages_dir = os.path.join(path, "stages")
        stages = []
        if os.path.isdir(stages_dir):
            entries = sorted(
                os.listdir(stages_dir), key=lambda s: int(s.split("_", 1)[0])
            )
            for e in entries:
                stages.append(load(os.path.join(stages_dir, e)))
        if simple == "Pipeline":
            inst = cls(stages=stages)
        else:
            inst = cls(stages)
        _apply_params(inst, meta)
        return inst

    data_dir = os.path.join(path, "data")
    data_rows = None
    if os.path.isdir(data_dir):
        data_rows = _read_parquet(data_dir).to_pylist()

    inst = _construct(cls, simple, data_rows, path, meta)
    _apply_params(inst, meta)
    return inst


def _construct(cls, simple: str, rows, path: str, meta):
    if simple == "StringIndexerModel":
        return cls(labelsArray=[list(l) for l in rows[0]["labelsArray"]])
    if simple == "OneHotEncoderModel":
        return cls(categorySizes=[int(v) for v in rows[0]["categorySizes"]])
    if simple == "StandardScalerModel":
        return cls(
            mean=torch.from_numpy(vector_from_struct(rows[0]["mean"])),
            std=torch.from_numpy(vector_from_struct(rows[0]["std"])),
        )
    if simple == "MinMaxScalerModel":
        return cls(
            originalMin=torch.from_numpy(vector_from_struct(rows[0]["originalMin"])),
            originalMax=torch.from_numpy(vector_from_struct(rows[0]["originalMax"])),
        )
    if simple == "VarianceThresholdSelectorModel":
        return cls(selectedFeatures=[int(i) for i in rows[0]["selectedFeatures"]])
    if simple == "UnivariateFeatureSelectorModel":
        return cls(selectedFeatures=[int(i) for i in rows[0]["selectedFeatures"]])
    if simple == "RobustScalerModel":
        return cls(
            median=torch.from_numpy(vector_from_struct(rows[0]["median"])),
            range_=torch.from_numpy(vector_from_struct(rows[0]["range"])),
        )
    if simple == "MaxAbsScalerModel":
        return cls(maxAbs=torch.from_numpy(vector_from_struct(rows[0]["maxAbs"])))
    if simple == "ImputerModel":
        return cls(surrogates=[float(v) for v in rows[0]["surrogates"]])
    if simple == "LogisticRegressionModel":
        r = rows[0]
        return cls(
            coefficients=torch.from_numpy(matrix_from_struct(r["coefficientMatrix"])).to(torch.float32),
            intercepts=torch.from_numpy(vector_from_struct(r["interceptVector"])).to(torch.float32),
            numClasses=int(r["numClasses"]),
            numFeatures=int(r["numFeatures"]),
            isMultinomial=bool(r["isMultinomial"]),
        )
    if simple == "NaiveBayesModel":
        r0 = rows[0]
        th = torch.from_numpy(matrix_from_struct(r0["theta"]))
        sg = torch.from_numpy(matrix_from_struct(r0["sigma"]))
        return cls(pi=torch.from_numpy(vector_from_struct(r0["pi"])), theta=th,
                   sigma=sg if sg.numel() else torch.zeros(0),
                   numClasses=th.shape[0], numFeatures=th.shape[1],
                   modelType=r0["modelType"])
    if simple == "BisectingKMeansModel":
        r0 = rows[0]
        return cls(centers=torch.from_numpy(matrix_from_struct(r0["clusterCenters"])),
                   trainingCost=float(r0["trainingCost"]))
    if simple == "GaussianMixtureModel":
        r0 = rows[0]
        mus = torch.stack([torch.from_numpy(vector_from_struct(v))
                           for v in r0["mus"]])
        covs = torch.stack([torch.from_numpy(matrix_from_struct(m))
                            for m in r0["sigmas"]])
        return cls(weights=torch.tensor(r0["weights"], dtype=torch.float64),
                   mus=mus, covs=covs)
    if simple == "KMeansModel":
        r0 = rows[0]
        return cls(centers=torch.from_numpy(matrix_from_struct(r0["clusterCenters"])),
                   trainingCost=float(r0["trainingCost"]))
    if simple == "PCAModel":
        r0 = rows[0]
        return cls(pc=torch.from_numpy(matrix_from_struct(r0["pc"])),
                   explainedVariance=torch.from_numpy(vector_from_struct(r0["explainedVariance"])))
    if simple == "IsotonicRegressionModel":
        r0 = rows[0]
        return cls(boundaries=torch.from_numpy(vector_from_struct(r0["boundaries"])),
                   predictions=torch.from_numpy(vector_from_struct(r0["predictions"])))
    if simple in ("FMClassificationModel", "FMRegressionModel"):
        r0 = rows[0]
        lin = torch.from_numpy(vector_from_struct(r0["linear"])).to(torch.float32)
        fac = torch.from_numpy(matrix_from_struct(r0["factors"])).to(torch.float32)
        return cls(intercept=float(r0["intercept"]), linear=lin, factors=fac,
                   numFeatures=int(lin.numel()))
    if simple == "LinearSVCModel":
        r0 = rows[0]
        coef = torch.from_numpy(vector_from_struct(r0["coefficients"])).to(torch.float32)
        return cls(coefficients=coef, intercept=float(r0["intercept"]),
                   numFeatures=int(coef.numel()))
    if simple == "GeneralizedLinearRegressionModel":
        r0 = rows[0]
        fam = meta.get("paramMap", {}).get("family", "gaussian")
        lnk = meta.get("paramMap", {}).get("link")
        if lnk is None:
            defaults = {"gaussian": "identity", "binomial": "logit",
                        "poisson": "log", "gamma": "inverse"}
            lnk = defaults.get(fam, "identity")
        return cls(
            coefficients=torch.from_numpy(vector_from_struct(r0["coefficients"])),
            intercept=float(r0["intercept"]), family=fam, link=lnk)
    if simple == "LinearRegressionModel":
        r0 = rows[0]
        coef = torch.from_numpy(vector_from_struct(r0["coefficients"])).to(torch.float32)
        return cls(coefficients=coef, intercept=float(r0["intercept"]),
                   numFeatures=int(coef.numel()))
    if simple == "VectorIndexerModel":
        nf = int(rows[0]["numFeatures"]) if rows else 0
        cmaps = {}
        for r0 in rows:
            if int(r0["featureIndex"]) >= 0:
                cmaps[int(r0["featureIndex"])] = {
                    float(v): i for i, v in enumerate(r0["categories"])
                }
        return cls(numFeatures=nf, categoryMaps=cmaps)
    if simple == "MultilayerPerceptronClassificationModel":
        layers = meta.get("paramMap", {}).get("layers") or meta.get(
            "defaultParamMap", {}
        ).get("layers")
        w = torch.from_numpy(vector_from_struct(rows[0]["weights"])).to(torch.float32)
        activation = meta.get("paramMap", {}).get("activation", "relu")
        return cls(layers=[int(x) for x in layers], weights=w, activation=activation)
    if simple == "RandomForestClassificationModel":
        trees = _trees_from_node_rows(rows, regression=False)
        n_classes = max(len(t.stats[0]) for t in trees)
        nf = _num_features_from_trees_meta(path, trees)
        return cls(trees=trees, numClasses=n_classes, numFeatures=nf)
    if simple == "GBTClassificationModel":
        # GBT classification trees are internal regression trees on
        # pseudo-residuals (variance impurity, (s, s2, n) stats).
        trees = _trees_from_node_rows(rows, regression=True)
        tm = _read_parquet(os.path.join(path, "treesMetadata")).to_pylist()
        weights = [r["weights"] for r in sorted(tm, key=lambda r: r["treeID"])]
        nf = _num_features_from_trees_meta(path, trees)
        return cls(trees=trees, treeWeights=weights, numFeatures=nf)
    if simple == "DecisionTreeClassificationModel":
        trees = _trees_from_node_rows(rows, regression=False)
        t = trees[0]
        return cls(tree=t, numClasses=len(t.stats[0]), numFeatures=0)
    if simple == "RandomForestRegressionModel":
        trees = _trees_from_node_rows(rows, regression=True)
        return cls(trees=trees, numFeatures=_num_features_from_trees_meta(path, trees))
    if simple == "GBTRegressionModel":
        trees = _trees_from_node_rows(rows, regression=True)
        tm = _read_parquet(os.path.join(path, "treesMetadata")).to_pylist()
        weights = [r["weights"] for r in sorted(tm, key=lambda r: r["treeID"])]
        return cls(trees=trees, treeWeights=weights,
                   numFeatures=_num_features_from_trees_meta(path, trees))
    if simple == "DecisionTreeRegressionModel":
        trees = _trees_from_node_rows(rows, regression=True)
        return cls(tree=trees[0],
                   numFeatures=_num_features_from_trees_meta(path, trees))
    # parameter-only stages
    return cls()


def _num_features_from_trees_meta(path: str, trees) -> int:
    tm_path = os.path.join(path, "treesMetadata")
    if os.path.isdir(tm_path):
        tm = _read_parquet(tm_path).to_pylist()
        if tm:
            try:
                return int(json.loads(tm[0]["metadata"]).get("numFeatures", 0))
            except Exception:  # noqa: BLE001
                pass
    mx = 0
    for t in trees:
        for f in t.feature:
            mx = max(mx, f + 1)
    return mx


# ---------------------------------------------------------------------------
# writer facade
# ---------------------------------------------------------------------------


class _WriterBase:
    def __init__(self, instance):
        self.instance = instance
        self._overwrite = False

    def overwrite(self):
        self._overwrite = True
        return self

    def save(self, path: str):
        if os.path.exists(path):
            if self._overwrite:
                import shutil

                shutil.rmtree(path)
            else:
                raise IOError(
                    f"Path {path} already exists. Use write().overwrite().save(path)."
                )
        _save_instance(self.instance, path)


class GenericWriter(_WriterBase):
    pass


class PipelineWriter(_WriterBase):
    pass


class PipelineModelWriter(_WriterBase):
    pass
