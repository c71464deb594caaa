#!/usr/bin/env python3
"""Author golden Spark-ML model directories INDEPENDENTLY of flowml's writer.

Written straight from the SURVEY.md §2.2/§3.5 schema spec with raw
json/pyarrow (no flowml imports), so ``flowml.io.load`` is tested against
files it did not produce. Run once; outputs are
committed under tests/fixtures/golden/.
"""

import json
import os
import shutil

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq

HERE = os.path.dirname(os.path.abspath(__file__))
OUT = os.path.join(HERE, "golden")


def write_meta(path, cls, uid, param_map, default_map):
    os.makedirs(os.path.join(path, "metadata"), exist_ok=True)
    meta = {
        "class": cls,
        "timestamp": 1726000000000,
        "sparkVersion": "3.5.1",
        "uid": uid,
        "paramMap": param_map,
        "defaultParamMap": default_map,
    }
    with open(os.path.join(path, "metadata", "part-00000"), "w") as f:
        f.write(json.dumps(meta) + "\n")
    open(os.path.join(path, "metadata", "_SUCCESS"), "w").close()


def write_data(path, rows, schema=None):
    d = os.path.join(path, "data")
    os.makedirs(d, exist_ok=True)
    t = pa.Table.from_pylist(rows, schema=schema)
    pq.write_table(t, os.path.join(d, "part-00000.parquet"))
    open(os.path.join(d, "_SUCCESS"), "w").close()


def dense_vec(vals):
    # VectorUDT: {type: 1 dense, size/indices null, values}
    return {"type": 1, "size": None, "indices": None,
            "values": [float(v) for v in vals]}


def dense_mat(m):
    # MatrixUDT: dense, COLUMN-major values, isTransposed false
    m = np.asarray(m, dtype=np.float64)
    return {"type": 1, "numRows": int(m.shape[0]), "numCols": int(m.shape[1]),
            "colPtrs": None, "rowIndices": None,
            "values": [float(v) for v in m.T.ravel()], "isTransposed": False}


def node(i, pred, imp, stats, gain, lc, rc, feat, lcot, ncat):
    return {"id": i, "prediction": float(pred), "impurity": float(imp),
            "impurityStats": [float(v) for v in stats], "gain": float(gain),
            "leftChild": lc, "rightChild": rc,
            "split": {"featureIndex": feat,
                      "leftCategoriesOrThreshold": [float(v) for v in lcot],
                      "numCategories": ncat}}


def main():
    if os.path.isdir(OUT):
        shutil.rmtree(OUT)

    # ---- LogisticRegressionModel (3 classes, 4 features) ------------------
    p = os.path.join(OUT, "lr")
    write_meta(p, "org.apache.spark.ml.classification.LogisticRegressionModel",
               "logreg_9f2b6c1a7d3e4f50",
               {"maxIter": 25, "regParam": 0.05, "elasticNetParam": 0.0},
               {"featuresCol": "features", "labelCol": "label",
                "predictionCol": "prediction", "maxIter": 100,
                "regParam": 0.0, "tol": 1e-06})
    W = np.array([[0.5, -1.25, 2.0, 0.125],
                  [-0.75, 0.25, 1.5, -2.5],
                  [3.0, -0.5, 0.0625, 1.0]])
    b = np.array([0.25, -0.125, 1.75])
    write_data(p, [{"numClasses": 3, "numFeatures": 4,
                    "interceptVector": dense_vec(b),
                    "coefficientMatrix": dense_mat(W),
                    "isMultinomial": True}])

    # ---- StringIndexerModel ----------------------------------------------
    p = os.path.join(OUT, "indexer")
    write_meta(p, "org.apache.spark.ml.feature.StringIndexerModel",
               "strIdx_0a1b2c3d4e5f6071",
               {"inputCol": "proto", "outputCol": "protoIndex",
                "handleInvalid": "keep"},
               {"handleInvalid": "error", "stringOrderType": "frequencyDesc"})
    write_data(p, [{"labelsArray": [["tcp", "udp", "icmp"]]}])

    # ---- RandomForestClassificationModel: 2 trees, 3 classes --------------
    # tree 0: root categorical split on feature 1 (arity 3, {0,2} left);
    #         left leaf class 1, right leaf class 2  — 3-wide impurityStats
    #         (the width-3 classification/regression ambiguity regression)
    # tree 1: root continuous split feature 0 at threshold 0.5
    p = os.path.join(OUT, "rf")
    write_meta(p, "org.apache.spark.ml.classification.RandomForestClassificationModel",
               "rfc_77aa88bb99cc00dd",
               {"numTrees": 2, "maxDepth": 1, "seed": 42},
               {"numTrees": 20, "maxDepth": 5, "maxBins": 32,
                "impurity": "gini", "featureSubsetStrategy": "auto"})
    rows = []
    rows.append({"treeID": 0, "nodeData": node(
        0, 1.0, 0.591836, [3, 16, 9], 0.26, 1, 2, 1, [0.0, 2.0], 3)})
    rows.append({"treeID": 0, "nodeData": node(
        1, 1.0, 0.18, [1, 14, 1], -1.0, -1, -1, -1, [], -1)})
    rows.append({"treeID": 0, "nodeData": node(
        2, 2.0, 0.42, [2, 2, 8], -1.0, -1, -1, -1, [], -1)})
    rows.append({"treeID": 1, "nodeData": node(
        0, 0.0, 0.56, [15, 8, 5], 0.2, 1, 2, 0, [0.5], 0)})
    rows.append({"treeID": 1, "nodeData": node(
        1, 0.0, 0.3, [12, 2, 1], -1.0, -1, -1, -1, [], -1)})
    rows.append({"treeID": 1, "nodeData": node(
        2, 1.0, 0.5, [3, 6, 4], -1.0, -1, -1, -1, [], -1)})
    write_data(p, rows)
    tm = os.path.join(p, "treesMetadata")
    os.makedirs(tm, exist_ok=True)
    tmeta = [{"treeID": t, "metadata": json.dumps({"numFeatures": 2}),
              "weights": 1.0} for t in range(2)]
    pq.write_table(pa.Table.from_pylist(tmeta),
                   os.path.join(tm, "part-00000.parquet"))
    open(os.path.join(tm, "_SUCCESS"), "w").close()

    # ---- MultilayerPerceptronClassificationModel (layers 3-4-2) -----------
    p = os.path.join(OUT, "mlp")
    nw = 3 * 4 + 4 + 4 * 2 + 2
    w = (np.arange(nw, dtype=np.float64) - 10.0) / 8.0
    write_meta(p, "org.apache.spark.ml.classification."
               "MultilayerPerceptronClassificationModel",
               "mlpc_1122334455667788",
               {"layers": [3, 4, 2], "maxIter": 50, "seed": 7},
               {"solver": "l-bfgs", "blockSize": 128, "tol": 1e-06})
    write_data(p, [{"weights": dense_vec(w)}])

    # ---- PipelineModel: StringIndexerModel -> LogisticRegressionModel ----
    p = os.path.join(OUT, "pipeline")
    write_meta(p, "org.apache.spark.ml.PipelineModel",
               "pipeline_aabbccddeeff0011", {}, {})
    s0 = os.path.join(p, "stages", "0_strIdx_0a1b2c3d4e5f6071")
    write_meta(s0, "org.apache.spark.ml.feature.StringIndexerModel",
               "strIdx_0a1b2c3d4e5f6071",
               {"inputCol": "proto", "outputCol": "protoIndex"},
               {"handleInvalid": "error"})
    write_data(s0, [{"labelsArray": [["a", "b"]]}])
    s1 = os.path.join(p, "stages", "1_logreg_9f2b6c1a7d3e4f50")
    write_meta(s1, "org.apache.spark.ml.classification.LogisticRegressionModel",
               "logreg_9f2b6c1a7d3e4f50", {"maxIter": 5}, {})
    write_data(s1, [{"numClasses": 2, "numFeatures": 2,
                     "interceptVector": dense_vec([0.0, 0.5]),
                     "coefficientMatrix": dense_mat([[1.0, -1.0], [0.5, 0.25]]),
                     "isMultinomial": True}])
    # ---- GBTClassificationModel: internal regression trees ---------------
    # width-3 impurityStats here are (sum, sum_sq, n) — the OTHER side of
    # the width-3 ambiguity; numCategories=0 continuous root, leaf preds
    # are means (s/n)
    p = os.path.join(OUT, "gbt")
    write_meta(p, "org.apache.spark.ml.classification.GBTClassificationModel",
               "gbtc_5566778899aabbcc",
               {"maxIter": 2, "maxDepth": 1, "stepSize": 0.1, "seed": 1},
               {"maxIter": 20, "lossType": "logistic"})
    rows = []
    rows.append({"treeID": 0, "nodeData": node(
        0, 0.125, 0.9375, [2.0, 4.5, 16.0], 0.5, 1, 2, 0, [1.5], 0)})
    rows.append({"treeID": 0, "nodeData": node(
        1, 0.75, 0.1, [6.0, 5.0, 8.0], -1.0, -1, -1, -1, [], -1)})
    rows.append({"treeID": 0, "nodeData": node(
        2, -0.5, 0.2, [-4.0, 2.5, 8.0], -1.0, -1, -1, -1, [], -1)})
    rows.append({"treeID": 1, "nodeData": node(
        0, 0.0625, 0.5, [1.0, 2.0, 16.0], 0.25, 1, 2, 1, [0.0, 2.0], 3)})
    rows.append({"treeID": 1, "nodeData": node(
        1, 0.25, 0.05, [2.5, 1.0, 10.0], -1.0, -1, -1, -1, [], -1)})
    rows.append({"treeID": 1, "nodeData": node(
        2, -0.25, 0.07, [-1.5, 1.0, 6.0], -1.0, -1, -1, -1, [], -1)})
    write_data(p, rows)
    tm = os.path.join(p, "treesMetadata")
    os.makedirs(tm, exist_ok=True)
    tmeta = [{"treeID": 0, "metadata": json.dumps({"numFeatures": 2}),
              "weights": 1.0},
             {"treeID": 1, "metadata": json.dumps({"numFeatures": 2}),
              "weights": 0.1}]
    pq.write_table(pa.Table.from_pylist(tmeta),
                   os.path.join(tm, "part-00000.parquet"))
    open(os.path.join(tm, "_SUCCESS"), "w").close()

    # ---- OneHotEncoderModel ----------------------------------------------
    p = os.path.join(OUT, "onehot")
    write_meta(p, "org.apache.spark.ml.feature.OneHotEncoderModel",
               "oneHot_99aabbccddeeff00",
               {"inputCols": ["pIdx", "sIdx"], "outputCols": ["pVec", "sVec"]},
               {"dropLast": True, "handleInvalid": "error"})
    write_data(p, [{"categorySizes": [3, 6]}])

    # ---- StandardScalerModel ---------------------------------------------
    p = os.path.join(OUT, "scaler")
    write_meta(p, "org.apache.spark.ml.feature.StandardScalerModel",
               "stdScal_1234abcd5678ef90",
               {"inputCol": "features", "outputCol": "scaled", "withMean": True},
               {"withMean": False, "withStd": True})
    write_data(p, [{"std": dense_vec([1.0, 2.0, 0.5]),
                    "mean": dense_vec([-1.0, 0.25, 3.0])}])

    # ---- GeneralizedLinearRegressionModel --------------------------------
    p = os.path.join(OUT, "glr")
    write_meta(p, "org.apache.spark.ml.regression.GeneralizedLinearRegressionModel",
               "glm_4455667788990011",
               {"family": "poisson", "link": "log", "maxIter": 10},
               {"family": "gaussian", "maxIter": 25, "tol": 1e-06})
    write_data(p, [{"intercept": 0.25,
                    "coefficients": dense_vec([0.5, -0.125, 2.0])}])

    # ---- GaussianMixtureModel --------------------------------------------
    p = os.path.join(OUT, "gmm")
    write_meta(p, "org.apache.spark.ml.clustering.GaussianMixtureModel",
               "GaussianMixture_aa11bb22cc33dd44",
               {"k": 2, "seed": 11},
               {"k": 2, "maxIter": 100, "tol": 0.01})
    write_data(p, [{
        "weights": [0.25, 0.75],
        "mus": [dense_vec([0.0, 1.0]), dense_vec([4.0, -2.0])],
        "sigmas": [dense_mat([[1.0, 0.1], [0.1, 2.0]]),
                   dense_mat([[0.5, 0.0], [0.0, 0.25]])],
    }])

    print(f"golden fixtures written under {OUT}")


if __name__ == "__main__":
    main()
