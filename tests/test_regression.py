"""Regression tree models + RegressionEvaluator."""

import numpy as np
import pytest
import torch

from flowml.evaluation import RegressionEvaluator
from flowml.frame import FlowFrame
from flowml.regression import (
    DecisionTreeRegressor,
    DecisionTreeRegressionModel,
    GBTRegressor,
    GBTRegressionModel,
    RandomForestRegressor,
    RandomForestRegressionModel,
)


def _reg_df(n=2000, seed=0):
    r = np.random.default_rng(seed)
    X = torch.from_numpy(r.normal(size=(n, 6)).astype(np.float32))
    y = (2.0 * X[:, 0] - X[:, 1] * X[:, 2] + 0.1 * torch.from_numpy(
        r.normal(size=n).astype(np.float32))).to(torch.float64)
    return FlowFrame({"features": X, "label": y}), X, y


class TestRegressionEvaluator:
    def test_metrics_vs_numpy(self):
        r = np.random.default_rng(3)
        lab = r.normal(size=200)
        pred = lab + r.normal(size=200) * 0.3
        df = FlowFrame({"label": torch.from_numpy(lab), "prediction": torch.from_numpy(pred)})
        err = lab - pred
        assert RegressionEvaluator(metricName="mse").evaluate(df) == pytest.approx((err ** 2).mean())
        assert RegressionEvaluator(metricName="rmse").evaluate(df) == pytest.approx(np.sqrt((err ** 2).mean()))
        assert RegressionEvaluator(metricName="mae").evaluate(df) == pytest.approx(np.abs(err).mean())
        r2 = 1 - (err ** 2).sum() / ((lab - lab.mean()) ** 2).sum()
        assert RegressionEvaluator(metricName="r2").evaluate(df) == pytest.approx(r2)
        assert RegressionEvaluator(metricName="rmse").isLargerBetter() is False
        assert RegressionEvaluator(metricName="r2").isLargerBetter() is True


class TestDecisionTreeRegressor:
    def test_fit_quality_vs_sklearn(self):
        from sklearn.tree import DecisionTreeRegressor as SkDT

        df, X, y = _reg_df()
        m = DecisionTreeRegressor(maxDepth=5, seed=0).fit(df)
        r2 = RegressionEvaluator(metricName="r2").evaluate(m.transform(df))
        sk = SkDT(max_depth=5).fit(X.numpy(), y.numpy())
        sk_r2 = sk.score(X.numpy(), y.numpy())
        assert r2 > sk_r2 - 0.1  # binned splits vs exact: close
        assert m.numNodes >= 3 and m.depth <= 5

    def test_leaf_value_is_mean(self):
        # single split on one feature: leaves predict the group means
        X = torch.tensor([[0.0], [0.0], [10.0], [10.0]])
        y = torch.tensor([1.0, 3.0, 10.0, 14.0], dtype=torch.float64)
        df = FlowFrame({"features": X, "label": y})
        m = DecisionTreeRegressor(maxDepth=1, maxBins=4).fit(df)
        p = m.transform(df)["prediction"].tolist()
        assert p[0] == p[1] == pytest.approx(2.0)
        assert p[2] == p[3] == pytest.approx(12.0)


class TestRandomForestRegressor:
    def test_fit_and_importances(self):
        df, X, y = _reg_df()
        m = RandomForestRegressor(numTrees=20, maxDepth=5, seed=1).fit(df)
        r2 = RegressionEvaluator(metricName="r2").evaluate(m.transform(df))
        assert r2 > 0.5
        imp = m.featureImportances.toArray()
        assert imp.shape == (6,) and imp[0] == imp.max()  # x0 dominates

    def test_persistence_roundtrip(self, tmp_path):
        df, _, _ = _reg_df(500)
        m = RandomForestRegressor(numTrees=4, maxDepth=3, seed=2).fit(df)
        m.save(str(tmp_path / "rfr"))
        lm = RandomForestRegressionModel.load(str(tmp_path / "rfr"))
        assert torch.allclose(lm.transform(df)["prediction"], m.transform(df)["prediction"])


class TestGBTRegressor:
    def test_monotone_training_rmse(self):
        df, _, _ = _reg_df()
        e = RegressionEvaluator(metricName="rmse")
        r5 = e.evaluate(GBTRegressor(maxIter=5, maxDepth=3, seed=0).fit(df).transform(df))
        r30 = e.evaluate(GBTRegressor(maxIter=30, maxDepth=3, seed=0).fit(df).transform(df))
        assert r30 < r5

    def test_beats_single_tree(self):
        df, _, _ = _reg_df()
        e = RegressionEvaluator(metricName="r2")
        gbt = e.evaluate(GBTRegressor(maxIter=30, maxDepth=3, seed=0).fit(df).transform(df))
        dt = e.evaluate(DecisionTreeRegressor(maxDepth=3, seed=0).fit(df).transform(df))
        assert gbt > dt

    def test_persistence_roundtrip(self, tmp_path):
        df, _, _ = _reg_df(500)
        m = GBTRegressor(maxIter=4, maxDepth=2, seed=1).fit(df)
        m.save(str(tmp_path / "gbtr"))
        lm = GBTRegressionModel.load(str(tmp_path / "gbtr"))
        assert lm.treeWeights == pytest.approx(m.treeWeights)
        assert torch.allclose(lm.transform(df)["prediction"], m.transform(df)["prediction"])


class TestLinearRegression:
    def test_recovers_coefficients(self):
        from flowml.regression import LinearRegression

        r = np.random.default_rng(0)
        X = torch.from_numpy(r.normal(size=(2000, 5)).astype(np.float32))
        w = torch.tensor([2.0, -1.0, 0.5, 0.0, 3.0])
        y = (X @ w + 1.5).to(torch.float64)
        m = LinearRegression(maxIter=100).fit(FlowFrame({"features": X, "label": y}))
        assert np.allclose(m.coefficients.toArray(), w.numpy(), atol=1e-3)
        assert m.intercept == pytest.approx(1.5, abs=1e-3)
        assert m.summary.r2 > 0.9999
        assert m.summary.totalIterations >= 1

    def test_l1_sparsifies(self):
        from flowml.regression import LinearRegression

        r = np.random.default_rng(1)
        X = torch.from_numpy(r.normal(size=(3000, 8)).astype(np.float32))
        y = (3.0 * X[:, 0] + 0.05 * torch.from_numpy(r.normal(size=3000).astype(np.float32))).to(torch.float64)
        m = LinearRegression(maxIter=150, regParam=0.3, elasticNetParam=1.0).fit(
            FlowFrame({"features": X, "label": y}))
        c = m.coefficients.toArray()
        assert abs(c[0]) > 1.0
        assert np.abs(c[1:]).max() < 1e-3  # L1 zeroed the noise features

    def test_weightcol_duplication(self):
        from flowml.regression import LinearRegression

        r = np.random.default_rng(2)
        X = torch.from_numpy(r.normal(size=(400, 3)).astype(np.float32))
        y = (X[:, 0] - X[:, 2]).to(torch.float64)
        w = torch.from_numpy(r.integers(1, 4, 400).astype(np.float32))
        df_w = FlowFrame({"features": X, "label": y, "w": w})
        idx = torch.repeat_interleave(torch.arange(400), w.to(torch.int64))
        df_d = FlowFrame({"features": X[idx], "label": y[idx]})
        mw = LinearRegression(maxIter=60, weightCol="w").fit(df_w)
        md = LinearRegression(maxIter=60).fit(df_d)
        assert np.allclose(mw.coefficients.toArray(), md.coefficients.toArray(), atol=1e-4)

    def test_persistence_roundtrip(self, tmp_path):
        from flowml.regression import LinearRegression, LinearRegressionModel

        r = np.random.default_rng(3)
        X = torch.from_numpy(r.normal(size=(300, 4)).astype(np.float32))
        y = (X[:, 1] * 2).to(torch.float64)
        m = LinearRegression(maxIter=40).fit(FlowFrame({"features": X, "label": y}))
        m.save(str(tmp_path / "linr"))
        lm = LinearRegressionModel.load(str(tmp_path / "linr"))
        assert np.allclose(lm.coefficients.toArray(), m.coefficients.toArray())
        assert lm.intercept == pytest.approx(m.intercept)
        assert not lm.hasSummary()


class TestFeatureImportancesEverywhere:
    def test_gbt_and_dt_importances(self):
        from flowml.classification import DecisionTreeClassifier, GBTClassifier
        from flowml.regression import DecisionTreeRegressor, GBTRegressor

        df, X, y = _reg_df(800)
        ycls = (y > y.median()).to(torch.float64)
        dfc = FlowFrame({"features": X, "label": ycls})
        for m in (
            GBTRegressor(maxIter=3, maxDepth=3, seed=0).fit(df),
            DecisionTreeRegressor(maxDepth=3, seed=0).fit(df),
            GBTClassifier(maxIter=3, maxDepth=3, seed=0).fit(dfc),
            DecisionTreeClassifier(maxDepth=3, seed=0).fit(dfc),
        ):
            imp = m.featureImportances.toArray()
            assert imp.shape == (6,)
            assert abs(imp.sum() - 1.0) < 1e-9 or imp.sum() == 0.0


class TestGBTRegressorWeightCol:
    def test_uniform_weight_equals_duplication(self):
        r = np.random.default_rng(10)
        X = torch.from_numpy(r.integers(0, 10, (400, 4)).astype(np.float32))
        y = (X[:, 0] * 2 - X[:, 1]).to(torch.float64)
        w = torch.full((400,), 2.0)
        df_w = FlowFrame({"features": X, "label": y, "w": w})
        idx = torch.repeat_interleave(torch.arange(400), 2)
        df_d = FlowFrame({"features": X[idx], "label": y[idx]})
        mw = GBTRegressor(maxIter=4, maxDepth=3, seed=0, weightCol="w").fit(df_w)
        md = GBTRegressor(maxIter=4, maxDepth=3, seed=0).fit(df_d)
        assert torch.equal(mw.transform(df_w)["prediction"],
                           md.transform(df_w)["prediction"])


class TestIsotonicRegression:
    def test_matches_sklearn(self, tmp_path):
        from sklearn.isotonic import IsotonicRegression as SkIso

        from flowml.regression import IsotonicRegression, IsotonicRegressionModel

        r = np.random.default_rng(0)
        x = np.sort(r.uniform(0, 10, 300)).astype(np.float32)
        y = (np.log1p(x) + r.normal(scale=0.2, size=300)).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(x),
                        "label": torch.from_numpy(y)})
        m = IsotonicRegression().fit(df)
        ours = m.transform(df)["prediction"].numpy()
        sk = SkIso(out_of_bounds="clip").fit(x, y)
        theirs = sk.predict(x)
        assert np.allclose(ours, theirs, atol=1e-6)
        # monotone
        assert (np.diff(ours) >= -1e-12).all()
        m.save(str(tmp_path / "iso"))
        lm = IsotonicRegressionModel.load(str(tmp_path / "iso"))
        assert np.allclose(lm.transform(df)["prediction"].numpy(), ours)

    def test_antitonic_and_vector_feature(self):
        from flowml.regression import IsotonicRegression

        r = np.random.default_rng(1)
        x = r.uniform(0, 5, 200).astype(np.float32)
        X = np.stack([r.normal(size=200).astype(np.float32), x], axis=1)
        y = (-2.0 * x + r.normal(scale=0.1, size=200)).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        m = IsotonicRegression(isotonic=False, featureIndex=1).fit(df)
        pred = m.transform(df)["prediction"].numpy()
        order = np.argsort(x)
        assert (np.diff(pred[order]) <= 1e-9).all()  # decreasing in x
        corr = np.corrcoef(pred, y)[0, 1]
        assert corr > 0.95


class TestExplainedVariance:
    def test_var_uses_label_mean(self):
        from flowml.evaluation import RegressionEvaluator

        lab = torch.tensor([0.0, 2.0, 4.0])
        pred = torch.tensor([1.0, 2.0, 3.0])
        df = FlowFrame({"label": lab, "prediction": pred})
        v = RegressionEvaluator(metricName="var").evaluate(df)
        ybar = float(lab.mean())
        expect = float(((pred - ybar) ** 2).mean())
        assert v == pytest.approx(expect)


class TestGeneralizedLinearRegression:
    def _xy(self, seed=0, n=3000):
        r = np.random.default_rng(seed)
        X = r.normal(size=(n, 4))
        beta = np.array([0.4, -0.3, 0.2, 0.1])
        return r, X, beta, X @ beta + 0.5

    def test_poisson_matches_sklearn(self):
        from sklearn.linear_model import PoissonRegressor

        from flowml.regression import GeneralizedLinearRegression

        r, X, beta, eta = self._xy()
        y = r.poisson(np.exp(eta)).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression(family="poisson").fit(df)
        sk = PoissonRegressor(alpha=0.0, max_iter=500, tol=1e-10).fit(X, y)
        assert np.allclose(m._coef.numpy(), sk.coef_, atol=2e-4)
        assert abs(m.intercept - sk.intercept_) < 2e-4
        pred = m.transform(df)["prediction"].numpy()
        assert np.allclose(pred, np.exp(X @ m._coef.numpy() + m.intercept),
                           rtol=1e-6)

    def test_binomial_matches_sklearn(self):
        from sklearn.linear_model import LogisticRegression as SkLR

        from flowml.regression import GeneralizedLinearRegression

        r, X, beta, eta = self._xy(1)
        y = (r.random(len(X)) < 1 / (1 + np.exp(-eta))).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression(family="binomial").fit(df)
        sk = SkLR(penalty=None, max_iter=500, tol=1e-10).fit(X, y)
        assert np.allclose(m._coef.numpy(), sk.coef_[0], atol=5e-4)

    def test_gamma_matches_sklearn(self):
        from sklearn.linear_model import GammaRegressor

        from flowml.regression import GeneralizedLinearRegression

        r, X, beta, eta = self._xy(2)
        y = r.gamma(shape=5.0, scale=np.exp(eta) / 5.0)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression(family="gamma", link="log").fit(df)
        sk = GammaRegressor(alpha=0.0, max_iter=1000, tol=1e-12).fit(X, y)
        assert np.allclose(m._coef.numpy(), sk.coef_, atol=1e-3)

    def test_gaussian_identity_equals_ols(self):
        from flowml.regression import GeneralizedLinearRegression

        r, X, beta, eta = self._xy(3)
        y = eta + 0.1 * r.normal(size=len(X))
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression().fit(df)
        Xa = np.column_stack([X, np.ones(len(X))])
        ols, *_ = np.linalg.lstsq(Xa, y, rcond=None)
        assert np.allclose(m._coef.numpy(), ols[:4], atol=1e-8)
        assert abs(m.intercept - ols[4]) < 1e-8

    def test_link_prediction_col_and_round_trip(self, tmp_path):
        from flowml.regression import (GeneralizedLinearRegression,
                                       GeneralizedLinearRegressionModel)

        r, X, beta, eta = self._xy(4)
        y = r.poisson(np.exp(eta)).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression(
            family="poisson", linkPredictionCol="eta").fit(df)
        out = m.transform(df)
        assert np.allclose(np.exp(out["eta"].numpy()),
                           out["prediction"].numpy(), rtol=1e-6)
        p = str(tmp_path / "glr")
        m.save(p)
        m2 = GeneralizedLinearRegressionModel.load(p)
        assert np.allclose(m2._coef.numpy(), m._coef.numpy())
        assert m2.getOrDefault(m2.family) == "poisson"
        assert torch.allclose(m.transform(df)["prediction"],
                              m2.transform(df)["prediction"])

    def test_unsupported_family(self):
        from flowml.regression import GeneralizedLinearRegression

        df = FlowFrame({"features": torch.zeros(4, 2), "label": np.ones(4)})
        with pytest.raises(ValueError, match="tweedie"):
            GeneralizedLinearRegression(family="tweedie").fit(df)


class TestGLRSummary:
    def test_gaussian_summary_matches_ols(self):
        import scipy.stats as st

        from flowml.regression import GeneralizedLinearRegression

        r = np.random.default_rng(0)
        X = r.normal(size=(500, 3))
        y = X @ np.array([1.0, -2.0, 0.0]) + 0.7 + r.normal(scale=1.5, size=500)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression().fit(df)
        s = m.summary
        Xa = np.column_stack([X, np.ones(500)])
        beta = np.concatenate([m._coef.numpy(), [m.intercept]])
        resid = y - Xa @ beta
        sigma2 = resid @ resid / 496
        se = np.sqrt(np.diag(sigma2 * np.linalg.inv(Xa.T @ Xa)))
        assert np.allclose(s.coefficientStandardErrors, se, rtol=1e-4)
        assert np.allclose(s.tValues, beta / se, rtol=1e-4)
        assert np.allclose(
            s.pValues, 2 * st.t.sf(np.abs(beta / se), 496),
            rtol=1e-3, atol=1e-12)
        assert s.pValues[2] > 0.05 and s.pValues[0] < 1e-6  # x2 is a null
        assert abs(s.dispersion - sigma2) < 1e-6
        assert s.residualDegreeOfFreedom == 496

    def test_poisson_summary_z_inference(self):
        from flowml.regression import GeneralizedLinearRegression

        r = np.random.default_rng(1)
        X = r.normal(size=(4000, 3))
        y = r.poisson(np.exp(X @ np.array([0.5, 0.0, -0.3]) + 0.2)).astype(
            np.float64)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression(family="poisson").fit(df)
        s = m.summary
        assert s.dispersion == 1.0  # fixed for poisson
        assert s.pValues[1] > 0.01      # null coefficient
        assert s.pValues[0] < 1e-10 and s.pValues[2] < 1e-10
        # +-2 SE covers the truth for the informative coefficients
        for i, truth in ((0, 0.5), (2, -0.3)):
            assert abs(m._coef[i].item() - truth) < 3 * s.coefficientStandardErrors[i]

    def test_summary_absent_after_load(self, tmp_path):
        from flowml.regression import (GeneralizedLinearRegression,
                                       GeneralizedLinearRegressionModel)

        r = np.random.default_rng(2)
        X = r.normal(size=(200, 2))
        y = X[:, 0] + r.normal(size=200)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = GeneralizedLinearRegression().fit(df)
        assert m.hasSummary()
        p = str(tmp_path / "g")
        m.save(p)
        m2 = GeneralizedLinearRegressionModel.load(p)
        assert not m2.hasSummary()
        with pytest.raises(RuntimeError, match="summary"):
            _ = m2.summary


class TestLinearRegressionInferenceStats:
    def test_ols_stats_match(self):
        import scipy.stats as st

        from flowml.regression import LinearRegression

        r = np.random.default_rng(0)
        X = r.normal(size=(500, 3))
        y = X @ np.array([1.0, -2.0, 0.0]) + 0.7 + r.normal(scale=1.5, size=500)
        df = FlowFrame({"features": torch.from_numpy(X.astype(np.float32)),
                        "label": y})
        m = LinearRegression(maxIter=200).fit(df)
        s = m.summary
        Xa = np.column_stack([X, np.ones(500)])
        ols, *_ = np.linalg.lstsq(Xa, y, rcond=None)
        resid = y - Xa @ ols
        se = np.sqrt(np.diag((resid @ resid / 496) * np.linalg.inv(Xa.T @ Xa)))
        assert np.allclose(s.coefficientStandardErrors, se, rtol=1e-2)
        assert s.degreesOfFreedom == 496
        assert s.pValues[2] > 0.05 and s.pValues[0] < 1e-6

    def test_regularized_raises_like_spark(self):
        from flowml.regression import LinearRegression

        r = np.random.default_rng(1)
        X = r.normal(size=(100, 2)).astype(np.float32)
        y = X[:, 0].astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        m = LinearRegression(regParam=0.1).fit(df)
        with pytest.raises(RuntimeError, match="Std. Error"):
            _ = m.summary.pValues


def test_regression_models_debug_strings():
    from flowml.regression import (DecisionTreeRegressor, GBTRegressor,
                                   RandomForestRegressor)

    r = np.random.default_rng(0)
    X = r.normal(size=(400, 3)).astype(np.float32)
    y = (X[:, 0] * 2).astype(np.float64)
    df = FlowFrame({"features": torch.from_numpy(X),
                    "label": torch.from_numpy(y)})
    for est, head in ((RandomForestRegressor(numTrees=2, maxDepth=2, seed=0),
                       "RandomForestRegressionModel"),
                      (DecisionTreeRegressor(maxDepth=2, seed=0),
                       "DecisionTreeRegressionModel"),
                      (GBTRegressor(maxIter=2, maxDepth=2, seed=0),
                       "GBTRegressionModel")):
        dbg = est.fit(df).toDebugString
        assert dbg.startswith(head)
        assert "Predict:" in dbg
