"""LogisticRegression: convergence, families, regularization."""

import numpy as np
import pytest
import torch

from flowml import FlowFrame
from flowml.classification import LogisticRegression
from flowml.evaluation import MulticlassClassificationEvaluator


def _binary_df(n=800, seed=0):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, 4)).astype(np.float32)
    w = np.array([2.0, -1.0, 0.5, 0.0])
    y = (X @ w + 0.3 * r.normal(size=n) > 0).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y}), X, y


def _multi_df(n=900, k=4, seed=1):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, 6)).astype(np.float32)
    W = r.normal(size=(6, k))
    y = (X @ W + 0.3 * r.normal(size=(n, k))).argmax(1).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y}), X, y


def test_binomial_separable_accuracy():
    df, X, y = _binary_df()
    m = LogisticRegression(maxIter=100).fit(df)
    assert not m.isMultinomial
    acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(m.transform(df))
    assert acc > 0.93


def test_binomial_matches_sklearn_unregularized():
    from sklearn.linear_model import LogisticRegression as SkLR

    df, X, y = _binary_df()
    m = LogisticRegression(maxIter=200, tol=1e-9).fit(df)
    sk = SkLR(C=1e8, max_iter=2000, tol=1e-10).fit(X, y)
    ours = m._W[0].numpy()
    theirs = sk.coef_[0]
    assert np.allclose(ours, theirs, rtol=0.05, atol=0.05)
    assert abs(m.intercept - sk.intercept_[0]) < 0.1


def test_multinomial_accuracy_and_shapes():
    df, X, y = _multi_df()
    m = LogisticRegression(maxIter=100).fit(df)
    assert m.isMultinomial
    assert m.numClasses == 4
    cm = m.coefficientMatrix
    assert (cm.numRows, cm.numCols) == (4, 6)
    assert len(m.interceptVector) == 4
    out = m.transform(df)
    assert out["probability"].shape == (len(df), 4)
    probs = out["probability"].sum(dim=1)
    assert torch.allclose(probs, torch.ones_like(probs), atol=1e-5)
    acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(out)
    assert acc > 0.85


def test_objective_history_decreasing():
    df, _, _ = _binary_df()
    m = LogisticRegression(maxIter=50).fit(df)
    h = m.summary.objectiveHistory
    assert len(h) >= 2
    assert h[-1] < h[0]
    assert all(h[i + 1] <= h[i] + 1e-9 for i in range(len(h) - 1))


def test_l2_regularization_shrinks():
    df, _, _ = _binary_df()
    m0 = LogisticRegression(maxIter=100).fit(df)
    m1 = LogisticRegression(maxIter=100, regParam=1.0).fit(df)
    assert float(m1._W.norm()) < float(m0._W.norm())


def test_l1_sparsity():
    df, X, y = _binary_df()
    m = LogisticRegression(maxIter=200, regParam=0.1, elasticNetParam=1.0).fit(df)
    coefs = m._W[0].numpy()
    # true weight of feature 3 is 0 -> L1 should zero it
    assert abs(coefs[3]) < 1e-6
    assert (np.abs(coefs) > 1e-6).sum() >= 2  # informative features survive


def test_elasticnet_matches_sklearn_loosely():
    from sklearn.linear_model import LogisticRegression as SkLR

    df, X, y = _binary_df()
    lam, alpha = 0.05, 0.5
    m = LogisticRegression(
        maxIter=300, regParam=lam, elasticNetParam=alpha, standardization=False, tol=1e-9
    ).fit(df)
    n = len(y)
    sk = SkLR(
        penalty="elasticnet", solver="saga", l1_ratio=alpha, C=1.0 / (n * lam) * n / n,
        max_iter=5000, tol=1e-10,
    )
    # sklearn objective: (1/n)Σloss + (1/(C n))||w|| splits -> C = 1/(n·lam)
    sk.C = 1.0 / (n * lam)
    sk.fit(X, y)
    assert np.allclose(m._W[0].numpy(), sk.coef_[0], rtol=0.15, atol=0.05)


def test_family_override():
    df, _, _ = _binary_df()
    m = LogisticRegression(maxIter=20, family="multinomial").fit(df)
    assert m.isMultinomial
    assert m.numClasses == 2
    assert m._W.shape == (2, 4)


def test_threshold():
    df, _, _ = _binary_df()
    m = LogisticRegression(maxIter=50).fit(df)
    m.set(m.threshold, 0.99)
    out = m.transform(df)
    # high threshold -> few positive predictions
    frac_pos = float(out["prediction"].mean())
    m.set(m.threshold, 0.01)
    frac_pos_low = float(m.transform(df)["prediction"].mean())
    assert frac_pos < frac_pos_low


def test_weight_col():
    r = np.random.default_rng(3)
    X = r.normal(size=(200, 2)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    w = np.where(y == 1, 10.0, 1.0)
    df = FlowFrame({"features": torch.from_numpy(X), "label": y, "w": w})
    m = LogisticRegression(maxIter=50, weightCol="w").fit(df)
    # upweighting positives shifts intercept up vs unweighted
    m0 = LogisticRegression(maxIter=50).fit(df)
    assert m.intercept > m0.intercept


def test_fit_intercept_false():
    df, _, _ = _binary_df()
    m = LogisticRegression(maxIter=50, fitIntercept=False).fit(df)
    assert m.intercept == 0.0


class TestBinarySummaryCurves:
    def test_roc_pr_threshold_curves(self):
        from flowml.evaluation import BinaryClassificationEvaluator

        r = np.random.default_rng(0)
        X = r.normal(size=(800, 4)).astype(np.float32)
        y = (X[:, 0] - X[:, 2] + 0.4 * r.normal(size=800) > 0).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        m = LogisticRegression(maxIter=60).fit(df)
        s = m.summary
        # AUC agrees with the evaluator on the same score
        ev = BinaryClassificationEvaluator(
            rawPredictionCol="probability").evaluate(m.transform(df))
        assert abs(s.areaUnderROC - ev) < 1e-9
        roc = s.roc
        assert float(roc["FPR"][0]) == 0.0 and float(roc["TPR"][-1]) == 1.0
        assert (roc["TPR"][1:] >= roc["TPR"][:-1]).all()  # monotone
        pr = s.pr
        assert pr.columns == ["recall", "precision"]
        f1 = s.fMeasureByThreshold()
        assert 0.7 < float(f1["metric"].max()) <= 1.0
        p_at = s.precisionByThreshold
        r_at = s.recallByThreshold
        assert len(p_at) == len(r_at) == len(f1)

    def test_multiclass_curves_raise(self):
        r = np.random.default_rng(1)
        X = r.normal(size=(300, 4)).astype(np.float32)
        y = r.integers(0, 3, 300).astype(np.float64)
        m = LogisticRegression(maxIter=5).fit(
            FlowFrame({"features": torch.from_numpy(X), "label": y}))
        with pytest.raises(RuntimeError, match="binary-only"):
            _ = m.summary.areaUnderROC
        # shared multiclass metrics still work
        assert 0.0 <= m.summary.accuracy <= 1.0


def test_linear_svc_summary_curves():
    from flowml.classification import LinearSVC

    r = np.random.default_rng(2)
    X = r.normal(size=(500, 4)).astype(np.float32)
    y = (X[:, 0] - X[:, 1] + 0.3 * r.normal(size=500) > 0).astype(np.float64)
    df = FlowFrame({"features": torch.from_numpy(X), "label": y})
    m = LinearSVC(maxIter=40).fit(df)
    s = m.summary
    assert s.areaUnderROC > 0.9
    assert 0.8 < s.accuracy <= 1.0
    assert len(s.precisionByThreshold) > 10


def test_negative_labels_clear_error():
    for est in ("LogisticRegression", "RandomForestClassifier",
                "MultilayerPerceptronClassifier"):
        import flowml.classification as C

        cls = getattr(C, est)
        kw = {"layers": [2, 4, 2]} if est.startswith("Multilayer") else {}
        with pytest.raises(ValueError, match="non-negative"):
            cls(**kw).fit(FlowFrame({"features": torch.zeros(4, 2),
                                     "label": np.array([-1.0, 0, 1, 0])}))
