"""Factorization machines: must learn pairwise interactions a linear model
cannot, gradients verified vs autograd, persistence round-trip."""

import numpy as np
import pytest
import torch

from flowml.classification import FMClassificationModel, FMClassifier
from flowml.fm import FMRegressionModel, FMRegressor, _fm_raw
from flowml.frame import FlowFrame


def _xor_df(n=2000, seed=0):
    """Pure interaction target: y = x0*x1 (zero linear signal)."""
    r = np.random.default_rng(seed)
    X = torch.from_numpy(r.choice([-1.0, 1.0], size=(n, 4)).astype(np.float32))
    y = (X[:, 0] * X[:, 1] > 0).to(torch.float64)
    return FlowFrame({"features": X, "label": y}), X, y


class TestGradients:
    def test_matches_autograd(self):
        torch.manual_seed(1)
        N, F, k = 40, 5, 3
        X = torch.randn(N, F)
        w0 = torch.randn(1, requires_grad=True)
        w = torch.randn(F, requires_grad=True)
        V = torch.randn(F, k, requires_grad=True)
        y = torch.randn(N)
        loss = 0.5 * ((_fm_raw(X, w0, w, V) - y) ** 2).sum()
        loss.backward()
        dl = (_fm_raw(X, w0.detach(), w.detach(), V.detach()) - y)
        xv = X @ V.detach()
        gV = X.T @ (dl.unsqueeze(1) * xv) - (X * X).T @ dl.unsqueeze(1) * V.detach()
        assert torch.allclose(dl.sum(), w0.grad[0], atol=1e-5)
        assert torch.allclose(dl @ X, w.grad, atol=1e-5)
        assert torch.allclose(gV, V.grad, atol=1e-4)


class TestFMClassifier:
    def test_learns_xor_interaction(self):
        df, X, y = _xor_df()
        fm = FMClassifier(factorSize=4, maxIter=200, stepSize=0.5, seed=1).fit(df)
        acc = float((fm.transform(df)["prediction"] == y).float().mean())
        assert acc > 0.95  # linear models are ~50% here
        # a factorSize=0 (linear) FM must NOT solve it
        lin = FMClassifier(factorSize=0, maxIter=200, stepSize=0.5, seed=1).fit(df)
        acc_lin = float((lin.transform(df)["prediction"] == y).float().mean())
        assert acc_lin < 0.65

    def test_probability_and_persistence(self, tmp_path):
        df, X, y = _xor_df(seed=2)
        fm = FMClassifier(factorSize=4, maxIter=100, stepSize=0.5).fit(df)
        out = fm.transform(df)
        p = out["probability"]
        assert torch.allclose(p.sum(dim=1), torch.ones(len(p), dtype=p.dtype))
        fm.save(str(tmp_path / "fmc"))
        lm = FMClassificationModel.load(str(tmp_path / "fmc"))
        assert torch.equal(lm.transform(df)["prediction"], out["prediction"])


class TestFMRegressor:
    def test_learns_quadratic_interaction(self):
        r = np.random.default_rng(3)
        X = torch.from_numpy(r.normal(size=(2000, 4)).astype(np.float32))
        y = (3.0 * X[:, 0] * X[:, 1] + X[:, 2]).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        fm = FMRegressor(factorSize=4, maxIter=300, stepSize=0.3, seed=4).fit(df)
        from flowml.evaluation import RegressionEvaluator

        r2 = RegressionEvaluator(metricName="r2").evaluate(fm.transform(df))
        assert r2 > 0.9
        from flowml.regression import LinearRegression

        lr2 = RegressionEvaluator(metricName="r2").evaluate(
            LinearRegression(maxIter=100).fit(df).transform(df))
        assert r2 > lr2 + 0.3  # interactions beyond the linear model

    def test_persistence(self, tmp_path):
        r = np.random.default_rng(5)
        X = torch.from_numpy(r.normal(size=(300, 3)).astype(np.float32))
        y = (X[:, 0] * X[:, 1]).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        m = FMRegressor(factorSize=3, maxIter=60, stepSize=0.3).fit(df)
        m.save(str(tmp_path / "fmr"))
        lm = FMRegressionModel.load(str(tmp_path / "fmr"))
        assert torch.allclose(lm.transform(df)["prediction"], m.transform(df)["prediction"])
