"""RandomForest / GBT property tests (gain-optimality,
oracle-quality, determinism)."""

import numpy as np
import pytest
import torch

from flowml import FlowFrame
from flowml.classification import (
    GBTClassifier,
    RandomForestClassifier,
)
from flowml.classification.tree import (
    GrowthConfig,
    find_split_thresholds,
    grow_trees,
    split_scan,
)
from flowml.evaluation import MulticlassClassificationEvaluator
from flowml import ops


def _multi_df(n=3000, f=8, k=3, seed=2):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, f)).astype(np.float32)
    W = r.normal(size=(f, k))
    y = (X @ W + 0.5 * r.normal(size=(n, k))).argmax(1).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y}), X, y


class TestSplitScan:
    def test_gain_optimal_single_feature(self):
        """A perfectly separable threshold must be found with full gini gain."""
        # one node, one feature, 4 bins; class 0 in bins 0-1, class 1 in bins 2-3
        hist = torch.zeros(1, 1, 4, 2)
        hist[0, 0, 0, 0] = 10
        hist[0, 0, 1, 0] = 5
        hist[0, 0, 2, 1] = 7
        hist[0, 0, 3, 1] = 8
        f, b, g, parent, left = split_scan(hist, "gini", 1, 0.0)
        assert int(f[0]) == 0
        assert int(b[0]) == 1  # split after bin 1
        p = 15 / 30
        expected_gain = 2 * p * (1 - p)  # parent gini, children pure
        assert float(g[0]) == pytest.approx(expected_gain)
        assert left[0].tolist() == [15.0, 0.0]

    def test_picks_best_feature(self):
        hist = torch.zeros(1, 2, 4, 2)
        # feature 0: useless (same distribution in all bins)
        hist[0, 0, :, 0] = 5
        hist[0, 0, :, 1] = 5
        # feature 1: perfect
        hist[0, 1, 0, 0] = 20
        hist[0, 1, 3, 1] = 20
        f, b, g, _, _ = split_scan(hist, "gini", 1, 0.0)
        assert int(f[0]) == 1

    def test_min_instances_respected(self):
        hist = torch.zeros(1, 1, 4, 2)
        hist[0, 0, 0, 0] = 1  # lone row in bin 0
        hist[0, 0, 3, 1] = 100
        f, b, g, _, _ = split_scan(hist, "gini", 5, 0.0)
        assert int(f[0]) == -1  # split would leave 1 < 5 rows on the left

    def test_pure_node_no_split(self):
        hist = torch.zeros(1, 1, 4, 2)
        hist[0, 0, 0, 0] = 10
        hist[0, 0, 3, 0] = 10
        f, _, g, _, _ = split_scan(hist, "gini", 1, 0.0)
        assert int(f[0]) == -1

    def test_variance_impurity(self):
        # regression stats (s, s2, n): values 0 in bins 0-1, 10 in bins 2-3
        hist = torch.zeros(1, 1, 4, 3)
        for b in (0, 1):
            hist[0, 0, b] = torch.tensor([0.0, 0.0, 5.0])
        for b in (2, 3):
            hist[0, 0, b] = torch.tensor([50.0, 500.0, 5.0])
        f, b, g, _, _ = split_scan(hist, "variance", 1, 0.0)
        assert int(f[0]) == 0
        assert int(b[0]) == 1
        assert float(g[0]) == pytest.approx(25.0)  # var drops from 25 to 0


class TestBinning:
    def test_thresholds_and_codes(self):
        X = torch.tensor([[0.0], [1.0], [2.0], [3.0], [4.0], [5.0], [6.0], [7.0]], dtype=torch.float32)
        thr = find_split_thresholds(X, max_bins=4, seed=0)
        assert thr.shape == (1, 3)
        codes = ops.bin_codes(X, thr)
        assert codes.dtype == torch.uint8
        # monotone non-decreasing codes with values spanning several bins
        c = codes[:, 0].tolist()
        assert c == sorted(c)
        assert len(set(c)) >= 3

    def test_constant_feature(self):
        X = torch.full((10, 1), 3.14)
        thr = find_split_thresholds(X, max_bins=8, seed=0)
        # at most one finite (useless) threshold; all rows land in bin 0
        assert int(torch.isfinite(thr).sum()) <= 1
        codes = ops.bin_codes(X, thr)
        assert codes.max() == 0

    def test_integer_feature_split_consistency(self):
        """Binned split rule must equal the raw-feature rule at inference,
        including x == threshold (strict-< codes)."""
        X = torch.tensor([[0.0], [1.0], [1.0], [2.0], [2.0], [2.0]] * 10)
        thr = find_split_thresholds(X, max_bins=4, seed=0)
        codes = ops.bin_codes(X, thr)
        for b in range(thr.shape[1]):
            t = float(thr[0, b])
            if not np.isfinite(t):
                continue
            left_binned = codes[:, 0] <= b
            left_raw = X[:, 0] <= t
            assert torch.equal(left_binned, left_raw), (b, t)


class TestRandomForest:
    def test_accuracy_vs_sklearn_oracle(self):
        from sklearn.ensemble import RandomForestClassifier as SkRF

        df, X, y = _multi_df()
        m = RandomForestClassifier(numTrees=20, maxDepth=6, seed=1).fit(df)
        acc_ev = MulticlassClassificationEvaluator(metricName="accuracy")
        ours = acc_ev.evaluate(m.transform(df))
        sk = SkRF(n_estimators=20, max_depth=6, max_features="sqrt", random_state=0).fit(X, y)
        theirs = sk.score(X, y)
        assert ours >= theirs - 0.05

    def test_deterministic_given_seed(self):
        df, _, _ = _multi_df(n=800)
        m1 = RandomForestClassifier(numTrees=5, maxDepth=4, seed=9).fit(df)
        m2 = RandomForestClassifier(numTrees=5, maxDepth=4, seed=9).fit(df)
        p1 = m1.transform(df)["prediction"]
        p2 = m2.transform(df)["prediction"]
        assert torch.equal(p1, p2)
        r1 = m1.transform(df)["rawPrediction"]
        r2 = m2.transform(df)["rawPrediction"]
        assert torch.equal(r1, r2)

    def test_model_shapes_and_probability(self):
        df, _, _ = _multi_df(n=500)
        m = RandomForestClassifier(numTrees=7, maxDepth=3, seed=2).fit(df)
        assert len(m._trees) == 7
        out = m.transform(df)
        prob = out["probability"]
        assert torch.allclose(prob.sum(dim=1), torch.ones(len(df)), atol=1e-5)
        imps = m.featureImportances.toArray()
        assert imps.sum() == pytest.approx(1.0)
        assert (imps >= 0).all()

    def test_depth_limit(self):
        df, _, _ = _multi_df(n=500)
        m = RandomForestClassifier(numTrees=3, maxDepth=2, seed=0).fit(df)
        for t in m._trees:
            assert t.depth() <= 2

    def test_single_class_data(self):
        df = FlowFrame(
            {
                "features": torch.randn(50, 3),
                "label": np.ones(50),
            }
        )
        m = RandomForestClassifier(numTrees=2, maxDepth=3, seed=0).fit(df)
        pred = m.transform(df)["prediction"]
        assert (pred == 1.0).all()

    def test_max_memory_chunking_equivalent(self):
        """Tiny maxMemoryInMB must not change results (only chunking)."""
        df, _, _ = _multi_df(n=600)
        m1 = RandomForestClassifier(numTrees=4, maxDepth=4, seed=3, maxMemoryInMB=256).fit(df)
        m2 = RandomForestClassifier(numTrees=4, maxDepth=4, seed=3, maxMemoryInMB=1).fit(df)
        assert torch.equal(m1.transform(df)["prediction"], m2.transform(df)["prediction"])


def _binary_df(n=2000, seed=4):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, 6)).astype(np.float32)
    y = ((X[:, 0] * X[:, 1] + X[:, 2]) > 0).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y}), X, y


class TestGBT:
    def test_accuracy_vs_sklearn_oracle(self):
        from sklearn.ensemble import GradientBoostingClassifier as SkGBT

        df, X, y = _binary_df()
        m = GBTClassifier(maxIter=30, maxDepth=3, seed=0).fit(df)
        acc = MulticlassClassificationEvaluator(metricName="accuracy")
        ours = acc.evaluate(m.transform(df))
        sk = SkGBT(n_estimators=30, max_depth=3, learning_rate=0.1, random_state=0).fit(X, y)
        assert ours >= sk.score(X, y) - 0.05

    def test_monotone_training_loss(self):
        df, X, y = _binary_df(n=800)
        accs = []
        for iters in (1, 5, 20):
            m = GBTClassifier(maxIter=iters, maxDepth=3, seed=0).fit(df)
            accs.append(
                MulticlassClassificationEvaluator(metricName="accuracy").evaluate(m.transform(df))
            )
        assert accs[-1] >= accs[0]

    def test_tree_weights(self):
        df, _, _ = _binary_df(n=400)
        m = GBTClassifier(maxIter=5, maxDepth=2, stepSize=0.3, seed=0).fit(df)
        assert m.treeWeights[0] == 1.0
        assert all(w == pytest.approx(0.3) for w in m.treeWeights[1:])

    def test_probability_and_raw(self):
        df, _, _ = _binary_df(n=400)
        m = GBTClassifier(maxIter=5, maxDepth=2, seed=0).fit(df)
        out = m.transform(df)
        raw = out["rawPrediction"]
        assert torch.allclose(raw[:, 0], -raw[:, 1])
        prob = out["probability"]
        assert torch.allclose(prob.sum(dim=1), torch.ones(len(df)), atol=1e-6)

    def test_multiclass_rejected(self):
        df = FlowFrame({"features": torch.randn(30, 2), "label": np.array([0.0, 1.0, 2.0] * 10)})
        with pytest.raises(ValueError):
            GBTClassifier(maxIter=2).fit(df)

    def test_validation_early_stop(self):
        r = np.random.default_rng(8)
        X = r.normal(size=(1200, 5)).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float64)
        val = (np.arange(1200) % 4 == 0)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y, "isVal": val})
        m = GBTClassifier(
            maxIter=50, maxDepth=2, seed=0, validationIndicatorCol="isVal"
        ).fit(df)
        assert len(m._trees) <= 50


class TestCheckpointResume:
    def test_gbt_checkpoint_resume_bitwise(self, tmp_path):
        """restart from checkpoint reproduces the model."""
        from flowml.utils.checkpoint import latest_gbt_checkpoint, load_gbt_checkpoint

        df, _, _ = _binary_df(n=500)
        ck = str(tmp_path / "ck")
        full = GBTClassifier(maxIter=10, maxDepth=2, seed=7, checkpointInterval=5)
        full.setCheckpointDir(ck)
        m_full = full.fit(df)
        ckpt = latest_gbt_checkpoint(ck)
        assert ckpt is not None
        init, rounds = load_gbt_checkpoint(ckpt)
        assert rounds in (5, 10)
        # resume from round 5 checkpoint
        init5, r5 = load_gbt_checkpoint(str(tmp_path / "ck" / "round_000005"))
        resumed = GBTClassifier(maxIter=10, maxDepth=2, seed=7)
        resumed.setInitModel(init5)
        m_res = resumed.fit(df)
        p_full = m_full.transform(df)["prediction"]
        p_res = m_res.transform(df)["prediction"]
        assert torch.equal(p_full, p_res)


class TestDecisionTree:
    def test_single_tree_fit(self):
        from flowml.classification import DecisionTreeClassifier

        df, X, y = _multi_df(n=1500)
        m = DecisionTreeClassifier(maxDepth=5, seed=0).fit(df)
        acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(m.transform(df))
        assert acc > 0.5
        assert m.depth <= 5
        assert m.numNodes >= 3

    def test_matches_one_tree_forest(self):
        from flowml.classification import DecisionTreeClassifier, RandomForestClassifier

        df, _, _ = _multi_df(n=800)
        dt = DecisionTreeClassifier(maxDepth=4, seed=5).fit(df)
        rf = RandomForestClassifier(
            numTrees=1, maxDepth=4, seed=5, bootstrap=False, featureSubsetStrategy="all"
        ).fit(df)
        assert torch.equal(dt.transform(df)["prediction"], rf.transform(df)["prediction"])


class TestGlobalHistMode:
    """Deep-level GLOBAL histogram mode (l_max=0 → HBM atomics, one pass)
    must be bitwise-equivalent to the LDS-chunked path. With
    featureSubsetStrategy='all' the chunk layout cannot influence the model,
    so forcing either mode must give identical predictions."""

    def _fit(self, monkeypatch, mode: str):
        import numpy as np
        from flowml.classification import RandomForestClassifier
        from flowml.frame import FlowFrame

        monkeypatch.setenv("FLOWML_HIST_GLOBAL", mode)
        r = np.random.default_rng(3)
        X = torch.from_numpy(r.normal(size=(4000, 10)).astype(np.float32))
        y = ((X[:, 0] + X[:, 1] * X[:, 2]) > 0).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        m = RandomForestClassifier(
            numTrees=8, maxDepth=7, seed=11, featureSubsetStrategy="all",
            maxMemoryInMB=1,
        ).fit(df)
        return m.transform(df)["prediction"]

    def test_global_equals_lds_chunked(self, monkeypatch):
        p_lds = self._fit(monkeypatch, "0")
        p_glob = self._fit(monkeypatch, "1")
        p_auto = self._fit(monkeypatch, "-1")
        assert torch.equal(p_lds, p_glob)
        assert torch.equal(p_lds, p_auto)


class TestWeightCol:
    def test_gbt_uniform_weights_equal_duplication(self):
        """Uniform weight w=2 must train the identical model as doubling
        every row: the quantile thresholds of a uniformly-replicated multiset
        are unchanged and 2·t is exact in fp32, so all histogram stats match
        bitwise and every split and leaf value is identical. (w=3 would NOT
        be bitwise — 3·t rounds in fp32; non-uniform weights additionally
        change the threshold sketch.)"""
        import numpy as np
        from flowml.classification import GBTClassifier
        from flowml.frame import FlowFrame

        r = np.random.default_rng(4)
        # integer-grid features: few distinct values per column, so the
        # equi-depth threshold sketch is duplication-invariant (continuous
        # data would shift quantile positions under replication)
        X = torch.from_numpy(r.integers(0, 10, (600, 6)).astype(np.float32))
        y = ((X[:, 0] + X[:, 1]) > 9).to(torch.float64)
        w = torch.full((600,), 2.0)
        df_w = FlowFrame({"features": X, "label": y, "w": w})
        idx = torch.repeat_interleave(torch.arange(600), 2)
        df_dup = FlowFrame({"features": X[idx], "label": y[idx]})
        m_w = GBTClassifier(maxIter=4, maxDepth=3, seed=0, weightCol="w").fit(df_w)
        m_d = GBTClassifier(maxIter=4, maxDepth=3, seed=0).fit(df_dup)
        p_w = m_w.transform(df_w)["prediction"]
        p_d = m_d.transform(df_w)["prediction"]
        assert torch.equal(p_w, p_d)

    def test_gbt_nonuniform_weights_change_fit(self):
        import numpy as np
        from flowml.classification import GBTClassifier
        from flowml.frame import FlowFrame

        r = np.random.default_rng(6)
        X = torch.from_numpy(r.normal(size=(600, 6)).astype(np.float32))
        y = ((X[:, 0] + X[:, 1]) > 0).to(torch.float64)
        w = torch.where(y > 0, torch.full((600,), 8.0), torch.ones(600))
        df = FlowFrame({"features": X, "label": y, "w": w})
        m_w = GBTClassifier(maxIter=4, maxDepth=3, seed=0, weightCol="w").fit(df)
        m_u = GBTClassifier(maxIter=4, maxDepth=3, seed=0).fit(df)
        assert not torch.equal(m_w.transform(df)["prediction"],
                               m_u.transform(df)["prediction"])

    def test_rf_weightcol_changes_fit(self):
        import numpy as np
        from flowml.classification import RandomForestClassifier
        from flowml.frame import FlowFrame

        r = np.random.default_rng(5)
        X = torch.from_numpy(r.normal(size=(500, 5)).astype(np.float32))
        y = (X[:, 0] > 0).to(torch.float64)
        w = torch.where(y > 0, torch.full((500,), 10.0), torch.ones(500))
        df = FlowFrame({"features": X, "label": y, "w": w})
        m = RandomForestClassifier(numTrees=5, maxDepth=3, seed=1, weightCol="w",
                                   bootstrap=False).fit(df)
        # upweighting class 1 must push recall for class 1 up
        pred = m.transform(df)["prediction"]
        rec1 = float(((pred == 1) & (y == 1)).sum() / (y == 1).sum())
        assert rec1 > 0.9


class TestDebugStringsAndIterEval:
    def test_to_debug_string(self):
        r = np.random.default_rng(0)
        X = r.normal(size=(800, 4)).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        m = RandomForestClassifier(numTrees=2, maxDepth=2, seed=0).fit(df)
        dbg = m.toDebugString
        assert dbg.startswith("RandomForestClassificationModel")
        assert "If (feature" in dbg and "Predict:" in dbg
        assert dbg.count("Tree ") == 2

    def test_categorical_debug_string(self):
        r = np.random.default_rng(1)
        cat = r.integers(0, 4, 800)
        y = np.isin(cat, [0, 2]).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=800)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: 4}})
        m = RandomForestClassifier(numTrees=1, maxDepth=1, seed=0,
                                   bootstrap=False).fit(df)
        dbg = m.toDebugString
        assert "in {" in dbg  # categorical split renders as set membership

    def test_gbt_evaluate_each_iteration(self):
        r = np.random.default_rng(2)
        X = r.normal(size=(2000, 4)).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        m = GBTClassifier(maxIter=6, maxDepth=2, seed=0).fit(df)
        losses = m.evaluateEachIteration(df)
        assert len(losses) == 6
        assert losses[-1] < losses[0]  # boosting reduces training loss
        with pytest.raises(ValueError):
            m.evaluateEachIteration(df, loss="squared")
