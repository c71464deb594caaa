"""Categorical split-sets + VectorIndexer.

Spec: spark:.../ml/tree/impl/RandomForest.scala binsToBestSplit categorical
branches, spark:.../feature/VectorIndexer.scala. Oracles are brute-force
subset enumerations computed here in numpy.
"""

import os

import numpy as np
import pytest
import torch

from flowml.frame import FlowFrame
from flowml.classification import (
    DecisionTreeClassifier,
    GBTClassifier,
    RandomForestClassifier,
)


def _gini_gain(counts_left, counts_right):
    def gini(c):
        n = c.sum()
        if n == 0:
            return 0.0
        p = c / n
        return 1.0 - (p * p).sum()
    nl, nr = counts_left.sum(), counts_right.sum()
    n = nl + nr
    parent = counts_left + counts_right
    return gini(parent) - (nl * gini(counts_left) + nr * gini(counts_right)) / n


def _brute_force_best_cat_gain(cat, y, k, n_classes):
    """Best gini gain over ALL category subsets (exhaustive oracle)."""
    counts = np.zeros((k, n_classes))
    for c, lab in zip(cat, y):
        counts[int(c), int(lab)] += 1
    best = -np.inf
    for m in range(1, 1 << (k - 1)):
        left = counts[[c for c in range(k) if (m >> c) & 1]].sum(axis=0)
        right = counts.sum(axis=0) - left
        if left.sum() < 1 or right.sum() < 1:
            continue
        best = max(best, _gini_gain(left, right))
    return best


class TestCategoricalSplits:
    def test_binary_ordered_is_optimal(self):
        """Binary classification: centroid-sort prefix scan (Breiman) finds
        the globally best subset — compare the fitted root gain against the
        exhaustive oracle."""
        r = np.random.default_rng(7)
        n, k = 5000, 6
        cat = r.integers(0, k, n)
        p1 = np.array([0.9, 0.1, 0.7, 0.2, 0.5, 0.05])[cat]
        y = (r.random(n) < p1).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: k}})
        m = DecisionTreeClassifier(maxDepth=1, seed=0).fit(df)
        t = m._tree
        assert t.feature[0] == 0 and t.cat_mask[0] != 0
        oracle = _brute_force_best_cat_gain(cat, y, k, 2)
        assert abs(t.gain[0] - oracle) < 1e-9, (t.gain[0], oracle)

    def test_multiclass_unordered_matches_bruteforce(self):
        """3-class, arity 4 (unordered regime at maxBins=32): the enumerated
        subset split equals the exhaustive oracle exactly."""
        r = np.random.default_rng(8)
        n, k, C = 6000, 4, 3
        cat = r.integers(0, k, n)
        probs = np.array([[0.8, 0.1, 0.1],
                          [0.1, 0.8, 0.1],
                          [0.1, 0.1, 0.8],
                          [0.4, 0.4, 0.2]])
        y = np.array([r.choice(C, p=probs[c]) for c in cat]).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: k}})
        m = DecisionTreeClassifier(maxDepth=1, seed=0).fit(df)
        t = m._tree
        assert t.feature[0] == 0 and t.cat_mask[0] != 0
        oracle = _brute_force_best_cat_gain(cat, y, k, C)
        assert abs(t.gain[0] - oracle) < 1e-9

    def test_multiclass_high_arity_ordered_heuristic(self):
        """Arity above the unordered cutoff: ordered impurity-centroid scan
        still produces a valid, routed categorical split."""
        from flowml.classification.tree import max_unordered_arity

        assert max_unordered_arity(32) == 5  # Spark formula at default bins
        r = np.random.default_rng(9)
        n, k, C = 8000, 8, 3
        cat = r.integers(0, k, n)
        y = (cat % 3).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: k}})
        # all categories are pure -> centroids tie -> prefix splits in id
        # order (exactly Spark's behavior); a worst-case spine needs depth
        # k-1 to isolate every category
        m = DecisionTreeClassifier(maxDepth=7, seed=0).fit(df)
        pred = m.transform(df)["prediction"].numpy()
        assert (pred == y).mean() > 0.99

    def test_regression_tree_categorical(self):
        """Variance trees: mean-centroid ordered scan is exact (Breiman)."""
        from flowml.regression import DecisionTreeRegressor

        r = np.random.default_rng(10)
        n, k = 5000, 5
        cat = r.integers(0, k, n)
        means = np.array([3.0, -1.0, 7.0, 0.0, -4.0])
        y = means[cat] + 0.01 * r.normal(size=n)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X),
                        "label": torch.from_numpy(y)}).withMeta(
            "features", {"categorical": {0: k}})
        m = DecisionTreeRegressor(maxDepth=4, seed=0).fit(df)
        pred = m.transform(df)["prediction"].numpy()
        # with depth 4 >= ceil(log2(5)) the tree isolates every category
        resid = np.abs(pred - y).mean()
        assert resid < 0.05, resid

    def test_gbt_categorical(self):
        r = np.random.default_rng(11)
        n, k = 4000, 4
        cat = r.integers(0, k, n)
        y = np.isin(cat, [1, 3]).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: k}})
        m = GBTClassifier(maxIter=5, maxDepth=2, seed=0).fit(df)
        pred = m.transform(df)["prediction"].numpy()
        assert (pred == y).mean() > 0.999
        assert any((t.cat_mask != 0).any() for t in m._trees)

    def test_rf_mixed_features_beats_ordinal(self):
        """Mixed categorical+continuous: declaring the categorical feature
        lifts depth-2 accuracy vs the ordinal treatment on an alternating
        category->class map."""
        r = np.random.default_rng(12)
        n, k = 8000, 6
        cat = r.integers(0, k, n)
        x1 = r.normal(size=n)
        y = ((cat % 2 == 0) ^ (x1 > 1.2)).astype(np.float64)
        X = np.column_stack([cat, x1, r.normal(size=n)]).astype(np.float32)
        base = FlowFrame({"features": torch.from_numpy(X), "label": y})
        dfc = base.withMeta("features", {"categorical": {0: k}})
        rf = RandomForestClassifier(numTrees=10, maxDepth=2, seed=1, bootstrap=False)
        acc_ord = (rf.fit(base).transform(base)["prediction"].numpy() == y).mean()
        acc_cat = (rf.fit(dfc).transform(dfc)["prediction"].numpy() == y).mean()
        assert acc_cat > acc_ord + 0.05, (acc_cat, acc_ord)

    def test_persistence_round_trip_categorical(self, tmp_path):
        import pyarrow.parquet as pq
        from flowml.classification import RandomForestClassificationModel

        r = np.random.default_rng(13)
        n, k = 3000, 5
        cat = r.integers(0, k, n)
        y = np.isin(cat, [0, 2, 4]).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: k}})
        m = RandomForestClassifier(numTrees=4, maxDepth=3, seed=0,
                                   bootstrap=False).fit(df)
        assert any((t.cat_mask != 0).any() for t in m._trees)
        p = str(tmp_path / "rfcat")
        m.save(p)
        # NodeData categorical split fields are populated per the Spark schema
        rows = pq.read_table(os.path.join(p, "data")).to_pylist()
        cat_nodes = [r0["nodeData"] for r0 in rows
                     if r0["nodeData"]["split"]["numCategories"] > 0]
        assert cat_nodes, "no categorical split was serialized"
        for nd in cat_nodes:
            sp = nd["split"]
            assert sp["numCategories"] == k
            cats = sp["leftCategoriesOrThreshold"]
            assert cats == sorted(cats) and all(0 <= c < k for c in cats)
        m2 = RandomForestClassificationModel.load(p)
        assert torch.equal(m.transform(df)["prediction"],
                           m2.transform(df)["prediction"])
        for t1, t2 in zip(m._trees, m2._trees):
            assert np.array_equal(t1.cat_mask, t2.cat_mask)
            assert np.array_equal(t1.num_cats, t2.num_cats)

    def test_arity_validation(self):
        r = np.random.default_rng(14)
        X = np.column_stack([r.integers(0, 40, 200), r.normal(size=200)]).astype(np.float32)
        y = r.integers(0, 2, 200).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: 40}})
        with pytest.raises(ValueError, match="arity"):
            RandomForestClassifier(numTrees=2, maxBins=32, seed=0).fit(df)

    def test_bad_category_values_raise(self):
        r = np.random.default_rng(15)
        X = np.column_stack([r.normal(size=200), r.normal(size=200)]).astype(np.float32)
        y = r.integers(0, 2, 200).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: 4}})
        with pytest.raises(ValueError, match="categorical feature 0"):
            RandomForestClassifier(numTrees=2, seed=0).fit(df)


class TestPipelineNominalPropagation:
    def test_stringindexer_assembler_to_tree(self):
        """The reference app path: StringIndexer -> VectorAssembler marks the
        slot nominal, and the tree fit picks up categorical semantics with
        no explicit declaration."""
        from flowml.feature import StringIndexer, VectorAssembler
        from flowml.pipeline import Pipeline

        r = np.random.default_rng(16)
        n = 5000
        proto = r.choice(["a", "b", "c", "d"], n)
        lab = {"a": 1.0, "b": 0.0, "c": 1.0, "d": 0.0}
        y = np.array([lab[p] for p in proto])
        df = FlowFrame({
            "proto": proto,
            "x": torch.from_numpy(r.normal(size=n).astype(np.float32)),
            "label": y,
        })
        si = StringIndexer(inputCol="proto", outputCol="pIdx")
        va = VectorAssembler(inputCols=["pIdx", "x"], outputCol="features")
        dt = DecisionTreeClassifier(maxDepth=1, seed=0)
        model = Pipeline(stages=[si, va, dt]).fit(df)
        out = model.transform(df)
        assert (out["prediction"].numpy() == y).mean() == 1.0  # depth-1 solves it
        tree = model.stages[-1]._tree
        assert tree.cat_mask[0] != 0 and tree.num_cats[0] == 4

    def test_assembler_meta_offsets(self):
        from flowml.feature import VectorAssembler

        n = 10
        df = FlowFrame({
            "a": torch.zeros(n),
            "v": torch.zeros(n, 3),
            "b": torch.zeros(n),
        })
        df = df.withMeta("a", {"nominal": 3})
        df = df.withMeta("v", {"categorical": {1: 5}})
        df = df.withMeta("b", {"nominal": 2})
        out = VectorAssembler(inputCols=["a", "v", "b"], outputCol="f").transform(df)
        assert out.meta("f") == {"categorical": {0: 3, 2: 5, 4: 2}}


class TestVectorIndexer:
    def _df(self):
        r = np.random.default_rng(17)
        X = np.column_stack([
            r.integers(0, 3, 400).astype(np.float64),
            r.normal(size=400),
            r.choice([0.0, 2.5, -1.0], 400),
        ]).astype(np.float32)
        return FlowFrame({"features": torch.from_numpy(X)}), X

    def test_fit_detects_and_zero_first(self):
        from flowml.feature import VectorIndexer

        df, X = self._df()
        m = VectorIndexer(inputCol="features", outputCol="idx",
                          maxCategories=4).fit(df)
        assert set(m.categoryMaps) == {0, 2}
        assert m.categoryMaps[2][0.0] == 0  # zero keeps index 0
        assert m.categoryMaps[2] == {0.0: 0, -1.0: 1, 2.5: 2}
        out = m.transform(df)
        assert out.meta("idx") == {"categorical": {0: 3, 2: 3}}
        ix = out["idx"].numpy()
        assert np.allclose(ix[:, 1], X[:, 1])  # continuous passthrough

    def test_handle_invalid(self):
        from flowml.feature import VectorIndexer, VectorIndexerModel

        df, X = self._df()
        m = VectorIndexer(inputCol="features", outputCol="idx",
                          maxCategories=4).fit(df)
        X2 = X.copy()
        X2[0, 0] = 99.0  # unseen category
        df2 = FlowFrame({"features": torch.from_numpy(X2)})
        with pytest.raises(ValueError, match="unseen"):
            m.transform(df2)
        m_keep = m.copy({m.handleInvalid: "keep"})
        out = m_keep.transform(df2)
        assert out["idx"][0, 0] == 3.0  # extra index = numCategories
        assert out.meta("idx")["categorical"][0] == 4
        m_skip = m.copy({m.handleInvalid: "skip"})
        assert m_skip.transform(df2).count() == len(X2) - 1

    def test_maxcategories_boundary(self):
        from flowml.feature import VectorIndexer

        n = 100
        X = np.column_stack([np.arange(n) % 5, np.arange(n) % 6]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = VectorIndexer(inputCol="features", outputCol="idx",
                          maxCategories=5).fit(df)
        assert 0 in m.categoryMaps and 1 not in m.categoryMaps

    def test_pipeline_with_trees(self):
        """VectorIndexer -> RF: the canonical Spark example-pipeline shape."""
        from flowml.feature import VectorIndexer
        from flowml.pipeline import Pipeline

        r = np.random.default_rng(18)
        n = 4000
        cat = r.integers(0, 4, n).astype(np.float64) * 10  # raw values 0,10,20,30
        y = np.isin(cat, [0.0, 20.0]).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        vi = VectorIndexer(inputCol="features", outputCol="indexed", maxCategories=10)
        dt = DecisionTreeClassifier(featuresCol="indexed", maxDepth=1, seed=0)
        model = Pipeline(stages=[vi, dt]).fit(df)
        out = model.transform(df)
        assert (out["prediction"].numpy() == y).mean() == 1.0


class TestGBTCategoricalPersistence:
    def test_gbt_round_trip_with_categorical_splits(self, tmp_path):
        """GBT trees are regression trees internally — categorical splits
        on them must persist (mean-centroid split-sets, width-3 stats AND
        numCategories>0 together)."""
        from flowml.classification import GBTClassificationModel

        r = np.random.default_rng(19)
        n, k = 3000, 4
        cat = r.integers(0, k, n)
        y = np.isin(cat, [1, 3]).astype(np.float64)
        X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
            "features", {"categorical": {0: k}})
        m = GBTClassifier(maxIter=4, maxDepth=2, seed=0).fit(df)
        assert any((t.cat_mask != 0).any() for t in m._trees)
        p = str(tmp_path / "gbtcat")
        m.save(p)
        m2 = GBTClassificationModel.load(p)
        assert all(t.regression for t in m2._trees)
        for t1, t2 in zip(m._trees, m2._trees):
            assert np.array_equal(t1.cat_mask, t2.cat_mask)
        assert torch.allclose(m.transform(df)["rawPrediction"],
                              m2.transform(df)["rawPrediction"], atol=1e-5)


# ---------------------------------------------------------------------------
# Property tests (hypothesis): fitted root split vs brute-force oracle
# ---------------------------------------------------------------------------

from hypothesis import given, settings, strategies as st


@settings(max_examples=30, deadline=None)
@given(
    k=st.integers(2, 6),
    n_classes=st.integers(2, 4),
    seed=st.integers(0, 10_000),
)
def test_root_categorical_gain_matches_oracle(k, n_classes, seed):
    """For every regime where Spark's search is exact (binary-classification
    centroid sort; unordered multiclass enumeration), the fitted root gain
    equals the exhaustive-subset oracle. In the ordered-multiclass heuristic
    regime the fitted gain never exceeds the oracle."""
    from flowml.classification.tree import max_unordered_arity

    r = np.random.default_rng(seed)
    n = 600
    cat = r.integers(0, k, n)
    # random class propensities per category so splits are non-trivial
    probs = r.dirichlet(np.ones(n_classes), size=k)
    y = np.array([r.choice(n_classes, p=probs[c]) for c in cat]).astype(np.float64)
    if len(np.unique(y)) < 2:
        return
    X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
    df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
        "features", {"categorical": {0: k}})
    m = DecisionTreeClassifier(maxDepth=1, seed=0).fit(df)
    t = m._tree
    if t.feature[0] != 0 or t.cat_mask[0] == 0:
        return  # the noise feature won (possible) — nothing to compare
    oracle = _brute_force_best_cat_gain(cat, y, k, n_classes)
    exact = (n_classes == 2) or (k <= max_unordered_arity(32))
    if exact:
        assert abs(t.gain[0] - oracle) < 1e-9, (t.gain[0], oracle)
    else:
        assert t.gain[0] <= oracle + 1e-9
    # the recorded split-set routing reproduces the recorded child counts
    left = np.array([(int(t.cat_mask[0]) >> int(c)) & 1 for c in cat], bool)
    nl = left.sum()
    assert nl == t.stats[int(t.left[0])].sum()


@settings(max_examples=15, deadline=None)
@given(seed=st.integers(0, 10_000))
def test_mixed_cat_cont_predictions_self_consistent(seed):
    """transform() on the training frame routes every row to a leaf whose
    majority class is the prediction (structural invariant, any tree)."""
    r = np.random.default_rng(seed)
    n, k = 400, 4
    cat = r.integers(0, k, n)
    x1 = r.normal(size=n)
    y = ((cat % 2 == 0) ^ (x1 > 0)).astype(np.float64)
    X = np.column_stack([cat, x1]).astype(np.float32)
    df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
        "features", {"categorical": {0: k}})
    m = DecisionTreeClassifier(maxDepth=3, seed=0).fit(df)
    t = m._tree
    pred = m.transform(df)["prediction"].numpy()
    # manual traversal oracle in python
    for i in range(0, n, 17):
        node = 0
        while t.feature[node] >= 0:
            f = t.feature[node]
            if t.cat_mask[node] != 0:
                go_left = (int(t.cat_mask[node]) >> int(X[i, f])) & 1 == 1
            else:
                go_left = X[i, f] <= t.thr_raw[node]
            node = int(t.left[node] if go_left else t.right[node])
        assert pred[i] == float(np.argmax(t.stats[node]))


def test_index_to_string_reads_label_metadata():
    """Spark parity: IndexToString with no labels param reads them from the
    input column's attributes (attached by StringIndexerModel)."""
    from flowml.feature import IndexToString, StringIndexer

    df = FlowFrame({"s": np.array(["b", "a", "b", "c"], dtype=object)})
    m = StringIndexer(inputCol="s", outputCol="i").fit(df)
    out = m.transform(df)
    assert out.meta("i")["labels"] == m.labels
    back = IndexToString(inputCol="i", outputCol="s2").transform(out)
    assert back["s2"].tolist() == ["b", "a", "b", "c"]
    with pytest.raises(ValueError, match="no labels"):
        IndexToString(inputCol="i", outputCol="s2").transform(
            FlowFrame({"i": out["i"]}))


def test_max_arity_63_mask_boundary():
    """Arity 63 (the int64-mask ceiling): splits using the top category
    (bit 62) must round-trip and route correctly."""
    r = np.random.default_rng(20)
    n, k = 6000, 63
    cat = r.integers(0, k, n)
    left_set = set(range(0, 63, 2))          # evens left (incl. 62)
    y = np.isin(cat, list(left_set)).astype(np.float64)
    X = np.column_stack([cat, r.normal(size=n)]).astype(np.float32)
    df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
        "features", {"categorical": {0: k}})
    m = DecisionTreeClassifier(maxDepth=1, maxBins=64, seed=0).fit(df)
    t = m._tree
    assert t.feature[0] == 0 and t.cat_mask[0] != 0
    mask = int(t.cat_mask[0])
    got_left = {c for c in range(63) if (mask >> c) & 1}
    # the split is the planted partition or its complement
    assert got_left in (left_set, set(range(63)) - left_set)
    pred = m.transform(df)["prediction"].numpy()
    assert (pred == y).mean() == 1.0
    # persistence round-trip keeps the full-width mask
    import tempfile, os as _os
    d = tempfile.mkdtemp()
    m.save(_os.path.join(d, "t"))
    from flowml.classification import DecisionTreeClassificationModel
    m2 = DecisionTreeClassificationModel.load(_os.path.join(d, "t"))
    assert int(m2._tree.cat_mask[0]) == mask


def test_arity_64_rejected():
    r = np.random.default_rng(21)
    X = np.column_stack([r.integers(0, 64, 500),
                         r.normal(size=500)]).astype(np.float32)
    y = r.integers(0, 2, 500).astype(np.float64)
    df = FlowFrame({"features": torch.from_numpy(X), "label": y}).withMeta(
        "features", {"categorical": {0: 64}})
    with pytest.raises(ValueError, match="arity"):
        DecisionTreeClassifier(maxBins=128, seed=0).fit(df)
