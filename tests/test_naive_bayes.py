"""NaiveBayes vs sklearn oracles + persistence."""

import numpy as np
import pytest
import torch

from flowml.classification import NaiveBayes, NaiveBayesModel
from flowml.frame import FlowFrame


class TestMultinomial:
    def test_matches_sklearn(self):
        from sklearn.naive_bayes import MultinomialNB

        r = np.random.default_rng(0)
        X = r.poisson(3.0, size=(500, 8)).astype(np.float32)
        y = r.integers(0, 3, 500).astype(np.float64)
        # make classes separable-ish
        X[y == 1, :3] += 4
        X[y == 2, 5:] += 4
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        m = NaiveBayes(smoothing=1.0).fit(df)
        sk = MultinomialNB(alpha=1.0).fit(X, y)
        ours = m.transform(df)["prediction"].numpy()
        theirs = sk.predict(X)
        assert (ours == theirs).mean() > 0.99
        # log class priors match
        assert np.allclose(m.pi.toArray(), sk.class_log_prior_, atol=1e-9)
        assert np.allclose(m.theta.toArray(), sk.feature_log_prob_, atol=1e-9)

    def test_probability_normalized(self):
        r = np.random.default_rng(1)
        X = r.poisson(2.0, size=(100, 5)).astype(np.float32)
        y = r.integers(0, 2, 100).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        out = NaiveBayes().fit(df).transform(df)
        p = out["probability"]
        assert torch.allclose(p.sum(dim=1), torch.ones(100, dtype=p.dtype))


class TestBernoulli:
    def test_matches_sklearn(self):
        from sklearn.naive_bayes import BernoulliNB

        r = np.random.default_rng(2)
        X = (r.random((400, 10)) < 0.3).astype(np.float32)
        y = r.integers(0, 2, 400).astype(np.float64)
        X[y == 1, :4] = (r.random((int((y == 1).sum()), 4)) < 0.8).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        m = NaiveBayes(modelType="bernoulli", smoothing=1.0).fit(df)
        sk = BernoulliNB(alpha=1.0).fit(X, y)
        assert (m.transform(df)["prediction"].numpy() == sk.predict(X)).mean() > 0.99

    def test_rejects_non_binary(self):
        X = torch.tensor([[0.0, 2.0]])
        df = FlowFrame({"features": X, "label": torch.tensor([0.0])})
        with pytest.raises(ValueError):
            NaiveBayes(modelType="bernoulli").fit(df)


class TestGaussian:
    def test_matches_sklearn(self):
        from sklearn.naive_bayes import GaussianNB

        r = np.random.default_rng(3)
        X = r.normal(size=(600, 6)).astype(np.float32)
        y = r.integers(0, 3, 600).astype(np.float64)
        X[y == 1] += 2.0
        X[y == 2, :3] -= 2.0
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        m = NaiveBayes(modelType="gaussian").fit(df)
        sk = GaussianNB().fit(X, y)
        assert (m.transform(df)["prediction"].numpy() == sk.predict(X)).mean() > 0.99


class TestPersistence:
    def test_roundtrip(self, tmp_path):
        r = np.random.default_rng(4)
        X = r.poisson(2.0, size=(200, 5)).astype(np.float32)
        y = r.integers(0, 2, 200).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        m = NaiveBayes().fit(df)
        m.save(str(tmp_path / "nb"))
        lm = NaiveBayesModel.load(str(tmp_path / "nb"))
        assert torch.equal(lm.transform(df)["prediction"], m.transform(df)["prediction"])

    def test_gaussian_roundtrip(self, tmp_path):
        r = np.random.default_rng(5)
        X = r.normal(size=(200, 4)).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float64)
        df = FlowFrame({"features": torch.from_numpy(X), "label": torch.from_numpy(y)})
        m = NaiveBayes(modelType="gaussian").fit(df)
        m.save(str(tmp_path / "gnb"))
        lm = NaiveBayesModel.load(str(tmp_path / "gnb"))
        assert torch.allclose(lm.transform(df)["probability"], m.transform(df)["probability"])


class TestOneVsRest:
    def test_matches_direct_multinomial_quality(self):
        from flowml.classification import LogisticRegression, OneVsRest
        from flowml.evaluation import MulticlassClassificationEvaluator

        r = np.random.default_rng(7)
        X = torch.from_numpy(r.normal(size=(600, 5)).astype(np.float32))
        y = torch.from_numpy(r.integers(0, 3, 600).astype(np.float64))
        X[y == 1, 0] += 2.5
        X[y == 2, 1] -= 2.5
        df = FlowFrame({"features": X, "label": y})
        ovr = OneVsRest(classifier=LogisticRegression(maxIter=30)).fit(df)
        acc_ovr = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(ovr.transform(df))
        direct = LogisticRegression(maxIter=30).fit(df)
        acc_dir = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(direct.transform(df))
        assert acc_ovr > 0.85
        assert abs(acc_ovr - acc_dir) < 0.1
        assert len(ovr.models) == 3

    def test_requires_classifier(self):
        from flowml.classification import OneVsRest

        df = FlowFrame({"features": torch.zeros(4, 2), "label": torch.zeros(4, dtype=torch.float64)})
        with pytest.raises(ValueError):
            OneVsRest().fit(df)


class TestLinearSVC:
    def test_separates_and_matches_sklearn_quality(self):
        from sklearn.svm import LinearSVC as SkSVC

        from flowml.classification import LinearSVC

        r = np.random.default_rng(11)
        X = torch.from_numpy(r.normal(size=(800, 6)).astype(np.float32))
        y = ((X[:, 0] + 0.5 * X[:, 1]) > 0).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        m = LinearSVC(maxIter=80, regParam=0.01).fit(df)
        ours = (m.transform(df)["prediction"].numpy() == y.numpy()).mean()
        sk = SkSVC(C=1.0, max_iter=5000).fit(X.numpy(), y.numpy())
        theirs = sk.score(X.numpy(), y.numpy())
        assert ours > theirs - 0.02
        assert ours > 0.95

    def test_raw_and_threshold(self):
        from flowml.classification import LinearSVC

        r = np.random.default_rng(12)
        X = torch.from_numpy(r.normal(size=(300, 4)).astype(np.float32))
        y = (X[:, 0] > 0).to(torch.float64)
        m = LinearSVC(maxIter=40).fit(FlowFrame({"features": X, "label": y}))
        out = m.transform(FlowFrame({"features": X, "label": y}))
        raw = out["rawPrediction"]
        assert raw.shape == (300, 2)
        assert torch.allclose(raw[:, 0], -raw[:, 1])
        assert "probability" not in out.columns  # plain Classifier, like Spark

    def test_persistence(self, tmp_path):
        from flowml.classification import LinearSVC, LinearSVCModel

        r = np.random.default_rng(13)
        X = torch.from_numpy(r.normal(size=(200, 3)).astype(np.float32))
        y = (X[:, 1] > 0).to(torch.float64)
        m = LinearSVC(maxIter=30).fit(FlowFrame({"features": X, "label": y}))
        m.save(str(tmp_path / "svc"))
        lm = LinearSVCModel.load(str(tmp_path / "svc"))
        assert np.allclose(lm.coefficients.toArray(), m.coefficients.toArray())
        assert lm.intercept == pytest.approx(m.intercept)

    def test_ovr_model_persistence(self, tmp_path):
        from flowml.classification import LinearSVC, OneVsRest, OneVsRestModel

        r = np.random.default_rng(20)
        X = torch.from_numpy(r.normal(size=(300, 4)).astype(np.float32))
        y = torch.from_numpy(r.integers(0, 3, 300).astype(np.float64))
        X[y == 1, 0] += 3
        X[y == 2, 1] -= 3
        df = FlowFrame({"features": X, "label": y})
        m = OneVsRest(classifier=LinearSVC(maxIter=20)).fit(df)
        m.save(str(tmp_path / "ovr"))
        lm = OneVsRestModel.load(str(tmp_path / "ovr"))
        assert len(lm.models) == 3
        assert torch.equal(lm.transform(df)["prediction"], m.transform(df)["prediction"])
