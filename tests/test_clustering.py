"""KMeans + ClusteringEvaluator vs sklearn oracles."""

import numpy as np
import pytest
import torch

from flowml.clustering import ClusteringEvaluator, KMeans, KMeansModel
from flowml.frame import FlowFrame


def _blobs(n=600, seed=0):
    r = np.random.default_rng(seed)
    centers = np.array([[0.0, 0.0], [6.0, 0.0], [0.0, 6.0]])
    X = np.vstack([c + r.normal(scale=0.6, size=(n // 3, 2)) for c in centers]).astype(np.float32)
    labels = np.repeat(np.arange(3), n // 3)
    return X, labels


class TestKMeans:
    def test_recovers_blobs(self):
        X, labels = _blobs()
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = KMeans(k=3, seed=1).fit(df)
        pred = m.transform(df)["prediction"].numpy().astype(int)
        # cluster/label agreement up to permutation: each true blob maps to
        # one dominant cluster
        for c in range(3):
            dominant = np.bincount(pred[labels == c], minlength=3).max()
            assert dominant > 0.95 * (labels == c).sum()
        assert len(m.clusterCenters()) == 3
        assert m.trainingCost < 600 * 2 * 0.6**2 * 3  # sane inertia

    def test_cost_close_to_sklearn(self):
        from sklearn.cluster import KMeans as SkKMeans

        X, _ = _blobs(seed=3)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = KMeans(k=3, seed=5, maxIter=30).fit(df)
        sk = SkKMeans(n_clusters=3, n_init=5).fit(X.astype(np.float64))
        assert m.computeCost(df) < sk.inertia_ * 1.05

    def test_random_init_and_predict(self):
        X, _ = _blobs(seed=4)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = KMeans(k=3, seed=2, initMode="random").fit(df)
        single = m.predict([0.1, 0.1])
        assert 0 <= single < 3

    def test_persistence(self, tmp_path):
        X, _ = _blobs(seed=6)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = KMeans(k=3, seed=7).fit(df)
        m.save(str(tmp_path / "km"))
        lm = KMeansModel.load(str(tmp_path / "km"))
        assert torch.equal(lm.transform(df)["prediction"], m.transform(df)["prediction"])
        assert lm.trainingCost == pytest.approx(m.trainingCost)


class TestClusteringEvaluator:
    def test_silhouette_vs_sklearn(self):
        from sklearn.metrics import silhouette_score

        X, labels = _blobs(seed=8)
        df = FlowFrame({"features": torch.from_numpy(X),
                        "prediction": torch.from_numpy(labels.astype(np.float64))})
        ours = ClusteringEvaluator().evaluate(df)
        # Spark/our silhouette uses SQUARED distances; sklearn uses
        # euclidean — not identical, but both should be high for clean blobs
        theirs = silhouette_score(X, labels)
        assert ours > 0.7 and theirs > 0.7
        assert ClusteringEvaluator().isLargerBetter()

    def test_bad_labels_score_worse(self):
        X, labels = _blobs(seed=9)
        r = np.random.default_rng(0)
        bad = r.integers(0, 3, len(labels)).astype(np.float64)
        good = ClusteringEvaluator().evaluate(FlowFrame(
            {"features": torch.from_numpy(X), "prediction": torch.from_numpy(labels.astype(np.float64))}))
        worse = ClusteringEvaluator().evaluate(FlowFrame(
            {"features": torch.from_numpy(X), "prediction": torch.from_numpy(bad)}))
        assert good > worse


class TestGaussianMixture:
    def _blobs(self, seed=0):
        r = np.random.default_rng(seed)
        A = r.normal(loc=[-4.0, 0.0], scale=0.5, size=(400, 2))
        B = r.normal(loc=[4.0, 3.0], scale=1.0, size=(600, 2))
        return np.vstack([A, B]).astype(np.float32)

    def test_separates_and_matches_sklearn(self):
        from sklearn.mixture import GaussianMixture as SkGMM

        from flowml.clustering import GaussianMixture

        X = self._blobs()
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = GaussianMixture(k=2, seed=3).fit(df)
        sk = SkGMM(n_components=2, random_state=0).fit(X.astype(np.float64))
        ours = sorted(
            (w, tuple(np.round(mu.toArray(), 1)))
            for w, (mu, _) in zip(m.weights, m.gaussians)
        )
        theirs = sorted(
            (w, tuple(np.round(mu, 1)))
            for w, mu in zip(sk.weights_, sk.means_)
        )
        for (w1, mu1), (w2, mu2) in zip(ours, theirs):
            assert abs(w1 - w2) < 0.02
            assert np.allclose(mu1, mu2, atol=0.3)
        prob = m.transform(df)["probability"]
        assert prob.shape == (1000, 2)
        assert torch.allclose(prob.sum(dim=1),
                              torch.ones(1000), atol=1e-5)

    def test_round_trip(self, tmp_path):
        from flowml.clustering import GaussianMixture, GaussianMixtureModel

        X = self._blobs(1)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = GaussianMixture(k=2, seed=5).fit(df)
        p = str(tmp_path / "gmm")
        m.save(p)
        m2 = GaussianMixtureModel.load(p)
        assert np.allclose(m2.weights, m.weights)
        assert torch.allclose(m2._mus, m._mus)
        assert torch.allclose(m2._covs, m._covs)
        assert torch.equal(m.transform(df)["prediction"],
                           m2.transform(df)["prediction"])


class TestBisectingKMeans:
    def _blobs(self):
        r = np.random.default_rng(2)
        blobs = [r.normal(loc=m, scale=0.3, size=(150, 2))
                 for m in ([-4, 0], [4, 0], [0, 4], [0, -4])]
        return np.vstack(blobs).astype(np.float32)

    def test_separates_blobs(self):
        from flowml.clustering import BisectingKMeans

        X = self._blobs()
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = BisectingKMeans(k=4, seed=1).fit(df)
        pred = m.transform(df)["prediction"].numpy()
        for b in range(4):
            seg = pred[b * 150:(b + 1) * 150]
            vals, cnts = np.unique(seg, return_counts=True)
            assert cnts.max() / len(seg) > 0.98
        assert len(set(pred[::150])) >= 3

    def test_min_divisible_cluster_size(self):
        from flowml.clustering import BisectingKMeans

        X = self._blobs()
        df = FlowFrame({"features": torch.from_numpy(X)})
        # whole dataset = 600 rows; forbidding division of clusters smaller
        # than 601 rows leaves a single split attempt impossible after root
        m = BisectingKMeans(k=4, seed=1, minDivisibleClusterSize=601.0).fit(df)
        assert len(m.clusterCenters()) == 1

    def test_round_trip(self, tmp_path):
        from flowml.clustering import BisectingKMeans, BisectingKMeansModel

        X = self._blobs()
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = BisectingKMeans(k=3, seed=4).fit(df)
        p = str(tmp_path / "bkm")
        m.save(p)
        m2 = BisectingKMeansModel.load(p)
        assert torch.allclose(m2._centers, m._centers)
        assert m2.trainingCost == m.trainingCost
        assert torch.equal(m.transform(df)["prediction"],
                           m2.transform(df)["prediction"])


def test_clustering_summaries():
    from flowml.clustering import BisectingKMeans, GaussianMixture, KMeans

    r = np.random.default_rng(3)
    X = np.vstack([r.normal(loc=m, scale=0.4, size=(100, 2))
                   for m in ([-3, 0], [3, 0], [0, 3])]).astype(np.float32)
    df = FlowFrame({"features": torch.from_numpy(X)})
    km = KMeans(k=3, seed=1).fit(df)
    s = km.summary
    assert sum(s.clusterSizes) == 300 and len(s.clusterSizes) == 3
    assert s.trainingCost > 0 and s.numIter >= 1
    gm = GaussianMixture(k=3, seed=1).fit(df)
    assert np.isfinite(gm.summary.logLikelihood)
    assert gm.summary.probabilityCol == "probability"
    bk = BisectingKMeans(k=3, seed=1).fit(df)
    assert sum(bk.summary.clusterSizes) == 300
    # absent after load
    import tempfile, os
    d = tempfile.mkdtemp()
    km.save(os.path.join(d, "km"))
    from flowml.clustering import KMeansModel
    km2 = KMeansModel.load(os.path.join(d, "km"))
    assert not km2.hasSummary()


def test_single_vector_predict():
    from flowml.clustering import BisectingKMeans, GaussianMixture, KMeans
    from flowml.linalg import DenseVector

    r = np.random.default_rng(4)
    X = np.vstack([r.normal(loc=m, scale=0.3, size=(100, 2))
                   for m in ([-3, 0], [3, 0])]).astype(np.float32)
    df = FlowFrame({"features": torch.from_numpy(X)})
    for est in (KMeans(k=2, seed=1), GaussianMixture(k=2, seed=1),
                BisectingKMeans(k=2, seed=1)):
        m = est.fit(df)
        assert m.predict(DenseVector([-3.0, 0.0])) != m.predict([3.0, 0.0])
