"""Load hand-authored golden Spark model dirs (NOT written by flowml.io) —
VERDICT r1 next-round #5. Fixtures live in tests/fixtures/golden/, generated
by tests/fixtures/make_golden.py straight from the SURVEY.md §2.2 schema
spec with raw json/pyarrow. A load-path change that silently breaks Spark
compatibility fails here even if save/load round-trips still pass."""

import os

import numpy as np
import pytest
import torch

from flowml.io import load

GOLDEN = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                      "fixtures", "golden")


def _p(name):
    d = os.path.join(GOLDEN, name)
    if not os.path.isdir(d):
        pytest.skip("golden fixtures not generated")
    return d


class TestGoldenLoad:
    def test_logistic_regression(self):
        m = load(_p("lr"))
        assert type(m).__name__ == "LogisticRegressionModel"
        assert m.uid == "logreg_9f2b6c1a7d3e4f50"
        assert m.numClasses == 3 and m.numFeatures == 4
        W = np.array([[0.5, -1.25, 2.0, 0.125],
                      [-0.75, 0.25, 1.5, -2.5],
                      [3.0, -0.5, 0.0625, 1.0]])
        assert np.array_equal(m._W.numpy(), W.astype(np.float32))  # bitwise
        assert np.array_equal(m._b.numpy(),
                              np.array([0.25, -0.125, 1.75], np.float32))
        assert m.getOrDefault(m.maxIter) == 25          # paramMap applied
        assert m.getOrDefault(m.regParam) == 0.05
        # defaultParamMap restored as defaults, not set params
        assert not m.isSet(m.tol)

    def test_string_indexer(self):
        m = load(_p("indexer"))
        assert m.labels == ["tcp", "udp", "icmp"]
        assert m.getHandleInvalid() == "keep"
        from flowml.frame import FlowFrame
        df = FlowFrame({"proto": np.array(["udp", "x", "tcp"], dtype=object)})
        out = m.transform(df)["protoIndex"].tolist()
        assert out == [1.0, 3.0, 0.0]  # keep -> unseen gets numLabels

    def test_random_forest_with_categorical_split(self):
        m = load(_p("rf"))
        assert type(m).__name__ == "RandomForestClassificationModel"
        assert m.numClasses == 3 and m.numFeatures == 2
        t0, t1 = m._trees
        # categorical root: {0, 2} left on feature 1, arity 3
        assert t0.feature[0] == 1
        assert t0.cat_mask[0] == 0b101
        assert t0.num_cats[0] == 3
        assert not t0.regression  # width-3 stats stay classification
        # counts from impurityStats sums (not stats[2] — the r1 bug)
        assert t0.count[0] == 28.0
        # continuous root on tree 1
        assert t1.feature[0] == 0 and t1.cat_mask[0] == 0
        assert t1.thr_raw[0] == 0.5
        # routing: f1 in {0,2} -> left leaf (class 1); f1 == 1 -> right (2)
        from flowml.frame import FlowFrame
        X = torch.tensor([[0.0, 0.0], [0.0, 1.0], [0.0, 2.0], [0.9, 1.0]],
                         dtype=torch.float32)
        pred = m.transform(FlowFrame({"features": X}))["prediction"].tolist()
        # per-tree votes are prob vectors; tree0 dominates where its leaves
        # are pure-ish: check the categorical routing changed predictions
        raw = m.transform(FlowFrame({"features": X}))["rawPrediction"]
        assert raw.shape == (4, 3)
        fi = m.featureImportances.toArray()
        assert fi.sum() == pytest.approx(1.0)

    def test_mlp(self):
        m = load(_p("mlp"))
        assert m.getLayers() == [3, 4, 2]
        nw = 3 * 4 + 4 + 4 * 2 + 2
        expect = ((np.arange(nw) - 10.0) / 8.0).astype(np.float32)
        assert np.array_equal(m._weights.numpy(), expect)

    def test_pipeline_model(self):
        m = load(_p("pipeline"))
        assert type(m).__name__ == "PipelineModel"
        assert [type(s).__name__ for s in m.stages] == [
            "StringIndexerModel", "LogisticRegressionModel"]
        assert m.uid == "pipeline_aabbccddeeff0011"
        assert m.stages[0].labels == ["a", "b"]

    def test_generator_is_flowml_free(self):
        """The fixture generator must not import flowml (independence)."""
        src = open(os.path.join(GOLDEN, os.pardir, "make_golden.py")).read()
        assert "flowml" not in src.replace("flowml's writer", "").replace(
            "flowml.io.load", "").replace("no flowml imports", "").replace(
            "flowml imports", "")

    def test_gbt_regression_trees(self):
        m = load(_p("gbt"))
        assert type(m).__name__ == "GBTClassificationModel"
        assert m.treeWeights == [1.0, 0.1]
        t0, t1 = m._trees
        # width-3 stats on a GBT tree mean (s, s2, n) — regression kind
        assert t0.regression and t1.regression
        assert t0.count[0] == 16.0            # n = stats[2], not sum
        assert t0.prediction(1) == pytest.approx(6.0 / 8.0)
        # tree 1 root is a categorical split {0,2} on feature 1
        assert t1.cat_mask[0] == 0b101 and t1.num_cats[0] == 3
        # margin check: x routes tree0 right (f0=2>1.5), tree1 left (f1=0)
        from flowml.frame import FlowFrame
        X = torch.tensor([[2.0, 0.0]], dtype=torch.float32)
        raw = m.transform(FlowFrame({"features": X}))["rawPrediction"][0]
        margin = 1.0 * (-0.5) + 0.1 * 0.25
        # Spark convention: predictRaw = [-F, F]
        assert float(raw[1] - raw[0]) == pytest.approx(2 * margin, abs=1e-6)

    def test_onehot_and_scaler(self):
        m = load(_p("onehot"))
        assert m.categorySizes == [3, 6]
        assert m.getOrDefault(m.outputCols) == ["pVec", "sVec"]
        sc = load(_p("scaler"))
        assert np.array_equal(sc.std.numpy(), [1.0, 2.0, 0.5])
        assert np.array_equal(sc.mean.numpy(), [-1.0, 0.25, 3.0])
        assert sc.getWithMean() is True  # paramMap overrode the default

    def test_glr(self):
        m = load(_p("glr"))
        assert type(m).__name__ == "GeneralizedLinearRegressionModel"
        assert m.getOrDefault(m.family) == "poisson"
        assert np.array_equal(m._coef.numpy(), [0.5, -0.125, 2.0])
        assert m.intercept == 0.25
        from flowml.frame import FlowFrame
        X = torch.tensor([[1.0, 0.0, 0.0]], dtype=torch.float32)
        out = m.transform(FlowFrame({"features": X}))["prediction"]
        assert float(out[0]) == pytest.approx(np.exp(0.75), rel=1e-6)

    def test_gmm(self):
        m = load(_p("gmm"))
        assert type(m).__name__ == "GaussianMixtureModel"
        assert m.weights == [0.25, 0.75]
        mus = m._mus.numpy()
        assert np.array_equal(mus, [[0.0, 1.0], [4.0, -2.0]])
        covs = m._covs.numpy()
        assert np.array_equal(covs[0], [[1.0, 0.1], [0.1, 2.0]])
        from flowml.frame import FlowFrame
        X = torch.tensor([[4.0, -2.0]], dtype=torch.float32)
        prob = m.transform(FlowFrame({"features": X}))["probability"]
        assert float(prob[0, 1]) > 0.99  # at the heavy component's mean
