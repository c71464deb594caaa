"""KMeans clustering + ClusteringEvaluator (pyspark.ml.clustering /
spark:.../ml/clustering/KMeans.scala, evaluation/ClusteringEvaluator.scala).

Lloyd iterations over the resident shard: assignments are a fused
distance-matrix argmin (|x|² − 2x·cᵀ + |c|², rocBLAS GEMM on GPU), the
per-cluster (count, Σx) statistics are computed with a deterministic one-hot
GEMM and all-reduced across ranks (SURVEY.md §2.5 C2 pattern), and every
rank applies the identical center update — the replicated-state design of
the tree/LR fits. Init: k-means|| (default, Spark's oversampling init) or
random. ClusteringEvaluator implements Spark's squared-euclidean silhouette
(per-cluster moments ⇒ O(N·k), not O(N²)).
"""

from __future__ import annotations

import math

from typing import List, Optional

import torch

from flowml import dist
from flowml.base import Estimator, Model
from flowml.frame import FlowFrame
from flowml.param import (
    HasFeaturesCol,
    HasMaxIter,
    HasPredictionCol,
    HasProbabilityCol,
    HasSeed,
    HasTol,
    HasWeightCol,
    Param,
    Params,
    TypeConverters,
)


class _KMeansParams(HasFeaturesCol, HasPredictionCol, HasMaxIter, HasSeed, HasTol, HasWeightCol):
    k: Param[int] = Param(
        "undefined", "k", "The number of clusters to create (> 1).", TypeConverters.toInt
    )
    initMode: Param[str] = Param(
        "undefined", "initMode", "The initialization algorithm: random | k-means||",
        TypeConverters.toString,
    )
    initSteps: Param[int] = Param(
        "undefined", "initSteps", "The number of steps for k-means|| (> 0).",
        TypeConverters.toInt,
    )


def _dist2(X: torch.Tensor, C: torch.Tensor) -> torch.Tensor:
    """N×k squared euclidean distances via GEMM (clamped at 0)."""
    x2 = (X * X).sum(dim=1, keepdim=True)
    c2 = (C * C).sum(dim=1)
    return (x2 - 2.0 * (X @ C.T) + c2).clamp_min(0.0)


class KMeans(_KMeansParams, Estimator):
    def __init__(self, **kwargs):
        super().__init__()
        self._setDefault(featuresCol="features", predictionCol="prediction",
                         k=2, maxIter=20, initMode="k-means||", initSteps=2,
                         tol=1e-4, seed=42)
        self._set(**kwargs)

    # -- k-means|| init (replicated across ranks: seeded identically) -------
    def _init_centers(self, X: torch.Tensor, k: int) -> torch.Tensor:
        seed = self.getOrDefault(self.seed)
        mode = self.getOrDefault(self.initMode)
        N = X.shape[0]
        g = torch.Generator(device=X.device if X.is_cuda else "cpu")
        g.manual_seed(seed & 0x7FFFFFFF)
        if mode == "random":
            idx = torch.randperm(N, generator=g, device=X.device)[:k]
            return X[idx].clone().to(torch.float64)
        if mode != "k-means||":
            raise ValueError(f"Unsupported initMode {mode!r}")
        # NOTE: single-shard draw per rank then rank-0 broadcast keeps all
        # ranks identical; candidates are capped so the final local k-means
        # (on candidates) is trivial.
        first = int(torch.randint(0, N, (1,), generator=g, device=X.device))
        centers = X[first : first + 1].to(torch.float64)
        steps = self.getOrDefault(self.initSteps)
        l = 2 * k  # Spark's oversampling factor ~2k per step
        for _ in range(steps):
            d2 = _dist2(X.to(torch.float64), centers).min(dim=1).values
            total = float(d2.sum().clamp_min(1e-300))
            probs = (l * d2 / total).clamp(0.0, 1.0)
            u = torch.rand(N, generator=g, device=X.device)
            picked = torch.nonzero(u < probs, as_tuple=False).flatten()
            if len(picked):
                centers = torch.cat([centers, X[picked].to(torch.float64)], dim=0)
        # weight candidates by assigned counts, then greedy k-means++ on the
        # (small) candidate set to get exactly k centers
        d2 = _dist2(X.to(torch.float64), centers)
        assign = d2.argmin(dim=1)
        w = torch.bincount(assign, minlength=len(centers)).to(torch.float64)
        cand = centers
        chosen = [int(w.argmax())]
        for _ in range(min(k, len(cand)) - 1):
            cd2 = _dist2(cand, cand[chosen]).min(dim=1).values * w
            cd2[chosen] = -1.0
            chosen.append(int(cd2.argmax()))
        C = cand[chosen]
        if len(C) < k:  # degenerate: pad with random rows
            idx = torch.randperm(N, generator=g, device=X.device)[: k - len(C)]
            C = torch.cat([C, X[idx].to(torch.float64)], dim=0)
        return C.clone()

    def _fit(self, dataset: FlowFrame) -> "KMeansModel":
        X = dataset.col(self.getFeaturesCol())
        if not isinstance(X, torch.Tensor) or X.dim() != 2:
            raise TypeError("KMeans features must be a vector column")
        Xd = X.to(torch.float64)
        N, F = Xd.shape
        k = self.getOrDefault(self.k)
        if k < 2:
            raise ValueError("k must be > 1")
        centers = self._init_centers(Xd, k)
        if dist.is_initialized() and dist.get_world_size() > 1:
            dist.broadcast_(centers)  # rank-0 init wins (shards differ)
        tol = self.getOrDefault(self.tol)
        cost = float("inf")
        it = 0
        for it in range(self.getMaxIter()):
            d2 = _dist2(Xd, centers)
            mind2, assign = d2.min(dim=1)
            onehot = torch.zeros(N, k, dtype=torch.float64, device=Xd.device)
            onehot[torch.arange(N, device=Xd.device), assign] = 1.0
            # deterministic per-cluster stats: GEMM, then one all_reduce
            buf = torch.cat([
                onehot.sum(dim=0),                    # k counts
                (onehot.T @ Xd).reshape(-1),          # k×F sums
                mind2.sum().reshape(1),               # cost
            ])
            dist.all_reduce_(buf)
            cnt = buf[:k]
            sums = buf[k : k + k * F].view(k, F)
            new_cost = float(buf[-1])
            nonempty = cnt > 0
            new_centers = torch.where(
                nonempty.unsqueeze(1), sums / cnt.clamp_min(1.0).unsqueeze(1), centers
            )
            shift = float(((new_centers - centers) ** 2).sum(dim=1).max())
            centers = new_centers
            if shift <= tol * tol:
                cost = new_cost
                break
            cost = new_cost
        model = KMeansModel(centers=centers.cpu(), trainingCost=cost,
                            numIter=it + 1)
        self._copyValues(model)
        model._set_parent(self)
        model._summary = ClusteringSummary(
            model, dataset, self.getOrDefault(self.predictionCol), k, it + 1,
            trainingCost=cost)
        return model


class KMeansModel(_KMeansParams, Model):
    def __init__(self, centers: Optional[torch.Tensor] = None,
                 trainingCost: float = 0.0, numIter: int = 0):
        super().__init__()
        self._summary = None
        self._centers = centers if centers is not None else torch.zeros(0, 0, dtype=torch.float64)
        self.trainingCost = trainingCost
        self.numIter = numIter

    @property
    def summary(self) -> "ClusteringSummary":
        if self._summary is None:
            raise RuntimeError("No training summary available (absent after load)")
        return self._summary

    def hasSummary(self) -> bool:
        return self._summary is not None

    def predict(self, features) -> int:
        """Cluster index for a single vector (spark KMeansModel.predict)."""
        v = torch.as_tensor(
            features.toArray() if hasattr(features, "toArray") else features,
            dtype=torch.float64).reshape(1, -1)
        return int(_dist2(v, self._centers.to(v.device)).argmin())

    def clusterCenters(self) -> List:
        return [c.numpy() for c in self._centers]

    def numClusters(self) -> int:
        return self._centers.shape[0]

    def predict(self, features) -> int:
        from flowml.classification.base import _as_row_tensor

        t = _as_row_tensor(features).to(torch.float64)
        return int(_dist2(t, self._centers.to(t.device)).argmin(dim=1)[0])

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        X = dataset.col(self.getFeaturesCol()).to(torch.float64)
        C = self._centers.to(X.device)
        assign = _dist2(X, C).argmin(dim=1)
        return dataset.withColumn(self.getOrDefault(self.predictionCol),
                                  assign.to(torch.float64))

    def computeCost(self, dataset: FlowFrame) -> float:
        X = dataset.col(self.getFeaturesCol()).to(torch.float64)
        return float(_dist2(X, self._centers.to(X.device)).min(dim=1).values.sum())

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


class ClusteringEvaluator(HasFeaturesCol, HasPredictionCol, Params):
    """Spark's squared-euclidean silhouette via per-cluster moments:
    for point x, a = mean ||x−y||² over own cluster = |x|² − 2x·μ_c + m_c
    (m_c = mean |y|² of the cluster); b = min over other clusters.
    s = (b − a)/max(a, b); metric = mean s (larger is better)."""

    metricName: Param[str] = Param(
        "undefined", "metricName", "metric name in evaluation (silhouette)",
        TypeConverters.toString,
    )

    def __init__(self, *, featuresCol: Optional[str] = None,
                 predictionCol: Optional[str] = None, metricName: Optional[str] = None):
        super().__init__()
        self._setDefault(featuresCol="features", predictionCol="prediction",
                         metricName="silhouette")
        self._set(featuresCol=featuresCol, predictionCol=predictionCol,
                  metricName=metricName)

    def isLargerBetter(self) -> bool:
        return True

    def evaluate(self, dataset: FlowFrame) -> float:
        if self.getOrDefault(self.metricName) != "silhouette":
            raise ValueError("Only metricName='silhouette' is supported")
        X = dataset.col(self.getFeaturesCol()).to(torch.float64)
        a = dataset.col(self.getOrDefault(self.predictionCol)).reshape(-1).to(torch.int64)
        N, F = X.shape
        k = int(a.max()) + 1
        if k < 2:
            raise ValueError("silhouette requires at least 2 clusters")
        onehot = torch.zeros(N, k, dtype=torch.float64, device=X.device)
        onehot[torch.arange(N, device=X.device), a] = 1.0
        cnt = onehot.sum(dim=0)                                  # k
        mu = (onehot.T @ X) / cnt.clamp_min(1.0).unsqueeze(1)    # k×F
        x2 = (X * X).sum(dim=1)
        m2 = (onehot.T @ x2.unsqueeze(1)).squeeze(1) / cnt.clamp_min(1.0)  # k
        # mean squared distance of x to cluster c: |x|² − 2 x·μ_c + m2_c
        d = x2.unsqueeze(1) - 2.0 * (X @ mu.T) + m2              # N×k
        own = d.gather(1, a.unsqueeze(1)).squeeze(1)
        d_other = d.scatter(1, a.unsqueeze(1), float("inf"))
        b = d_other.min(dim=1).values
        s = torch.where(
            cnt[a] > 1,
            (b - own) / torch.maximum(own, b).clamp_min(1e-300),
            torch.zeros_like(own),
        )
        return float(s.mean())


class ClusteringSummary:
    """spark ClusteringSummary subset: lazy predictions + clusterSizes +
    numIter (KMeans adds trainingCost, GaussianMixture adds logLikelihood)."""

    def __init__(self, model, dataset, predictionCol: str, k: int,
                 numIter: int, **extra):
        self._model = model
        self._dataset = dataset
        self.predictionCol = predictionCol
        self.k = int(k)
        self.numIter = int(numIter)
        for kk, v in extra.items():
            setattr(self, kk, v)
        self._pred = None

    @property
    def predictions(self):
        if self._pred is None:
            self._pred = self._model.transform(self._dataset)
        return self._pred

    @property
    def clusterSizes(self):
        pred = self.predictions.col(self.predictionCol).to(torch.int64)
        sizes = torch.bincount(pred, minlength=self.k)
        from flowml import dist as _d

        sizes = sizes.to(torch.float64)
        _d.all_reduce_(sizes)
        return [int(v) for v in sizes]


# ---------------------------------------------------------------------------
# BisectingKMeans (spark:.../ml/clustering/BisectingKMeans.scala)
# ---------------------------------------------------------------------------


class _BisectingKMeansParams(HasFeaturesCol, HasPredictionCol, HasMaxIter,
                             HasSeed, HasWeightCol):
    k: Param[int] = Param(
        "undefined", "k", "The desired number of leaf clusters (> 1).",
        TypeConverters.toInt)
    minDivisibleClusterSize: Param[float] = Param(
        "undefined", "minDivisibleClusterSize",
        "The minimum number of points (if >= 1.0) or the minimum proportion "
        "of points (if < 1.0) of a divisible cluster.",
        TypeConverters.toFloat)

    def __init__(self):
        super().__init__()
        self._setDefault(featuresCol="features", predictionCol="prediction",
                         k=4, maxIter=20, minDivisibleClusterSize=1.0, seed=42)


class BisectingKMeans(_BisectingKMeansParams, Estimator):
    """Divisive hierarchical clustering: repeatedly bisect the largest
    divisible cluster with an inner 2-means until ``k`` leaf clusters exist
    (spark:.../ml/clustering/BisectingKMeans.scala: defaults k=4,
    maxIter=20, minDivisibleClusterSize=1.0). Replicated across ranks: the
    inner Lloyd statistics are all-reduced and the seeded split init is
    rank-0-broadcast, so every rank holds the identical tree of centers."""

    def __init__(self, **kwargs):
        super().__init__()
        self._set(**kwargs)

    def setK(self, v: int):
        return self.set(self.k, v)

    def _bisect(self, X: torch.Tensor, seed: int):
        """2-means on the rows of X (may be empty on some ranks)."""
        N = X.shape[0]
        F = X.shape[1]
        g = torch.Generator(device=X.device if X.is_cuda else "cpu")
        g.manual_seed(seed & 0x7FFFFFFF)
        if N >= 2:
            idx = torch.randperm(N, generator=g, device=X.device)[:2]
            centers = X[idx].clone()
        else:
            centers = torch.zeros(2, F, dtype=torch.float64, device=X.device)
        dist.broadcast_(centers)
        for _ in range(self.getMaxIter()):
            if N:
                d2 = _dist2(X, centers)
                assign = d2.argmin(dim=1)
                onehot = torch.zeros(N, 2, dtype=torch.float64, device=X.device)
                onehot[torch.arange(N, device=X.device), assign] = 1.0
                buf = torch.cat([onehot.sum(dim=0), (onehot.T @ X).reshape(-1)])
            else:
                buf = torch.zeros(2 + 2 * F, dtype=torch.float64, device=X.device)
            dist.all_reduce_(buf)
            cnt = buf[:2]
            sums = buf[2:].view(2, F)
            new_centers = torch.where(
                (cnt > 0).unsqueeze(1), sums / cnt.clamp_min(1.0).unsqueeze(1),
                centers)
            if torch.allclose(new_centers, centers, atol=1e-12):
                centers = new_centers
                break
            centers = new_centers
        return centers

    def _fit(self, dataset: FlowFrame) -> "BisectingKMeansModel":
        X = dataset.col(self.getFeaturesCol())
        if not isinstance(X, torch.Tensor) or X.dim() != 2:
            raise TypeError("BisectingKMeans features must be a vector column")
        Xd = X.to(torch.float64)
        N, F = Xd.shape
        k = self.getOrDefault(self.k)
        if k < 2:
            raise ValueError("k must be > 1")
        seed = self.getOrDefault(self.seed)
        mdcs = self.getOrDefault(self.minDivisibleClusterSize)
        n_tot = torch.tensor([float(N)], dtype=torch.float64, device=Xd.device)
        dist.all_reduce_(n_tot)
        min_size = mdcs if mdcs >= 1.0 else mdcs * float(n_tot)
        assign = torch.zeros(N, dtype=torch.int64, device=Xd.device)
        centers = [Xd.sum(dim=0)]
        c0 = torch.cat([centers[0], torch.tensor([float(N)], dtype=torch.float64,
                                                 device=Xd.device)])
        dist.all_reduce_(c0)
        centers = [c0[:F] / c0[F].clamp_min(1.0)]
        sizes = [float(c0[F])]
        divisible = [True]
        while len(centers) < k and any(
            d and s >= max(min_size, 2.0) for d, s in zip(divisible, sizes)
        ):
            # split the largest divisible cluster (global sizes — identical
            # on every rank, so the choice is replicated)
            cand = [i for i, (d, sz) in enumerate(zip(divisible, sizes))
                    if d and sz >= max(min_size, 2.0)]
            target = max(cand, key=lambda i: sizes[i])
            mask = assign == target
            sub = Xd[mask]
            cpair = self._bisect(sub, seed + 7919 * len(centers))
            if N:
                d2 = _dist2(sub, cpair)
                side = d2.argmin(dim=1)
            # new sizes, all-reduced
            sz = torch.zeros(2, dtype=torch.float64, device=Xd.device)
            if N and len(sub):
                sz[0] = float((side == 0).sum())
                sz[1] = float((side == 1).sum())
            dist.all_reduce_(sz)
            if float(sz.min()) == 0.0:
                divisible[target] = False  # degenerate split: mark leaf
                continue
            new_id = len(centers)
            if N and len(sub):
                sel = torch.nonzero(mask, as_tuple=False).flatten()
                assign[sel[side == 1]] = new_id
            centers[target] = cpair[0]
            centers.append(cpair[1])
            sizes[target] = float(sz[0])
            sizes.append(float(sz[1]))
            divisible[target] = True
            divisible.append(True)
        C = torch.stack(centers)
        # training cost: sum of squared distances to the final centers
        if N:
            cost_t = _dist2(Xd, C).min(dim=1).values.sum().reshape(1)
        else:
            cost_t = torch.zeros(1, dtype=torch.float64, device=Xd.device)
        dist.all_reduce_(cost_t)
        model = BisectingKMeansModel(centers=C.cpu(),
                                     trainingCost=float(cost_t))
        self._copyValues(model)
        model._set_parent(self)
        model._summary = ClusteringSummary(
            model, dataset, self.getOrDefault(self.predictionCol),
            len(centers), 0, trainingCost=float(cost_t))
        return model


class BisectingKMeansModel(_BisectingKMeansParams, Model):
    def __init__(self, centers: Optional[torch.Tensor] = None,
                 trainingCost: float = 0.0):
        super().__init__()
        self._summary = None
        self._centers = centers if centers is not None else torch.zeros(
            0, 0, dtype=torch.float64)
        self.trainingCost = float(trainingCost)

    @property
    def summary(self) -> "ClusteringSummary":
        if self._summary is None:
            raise RuntimeError("No training summary available (absent after load)")
        return self._summary

    def hasSummary(self) -> bool:
        return self._summary is not None

    def predict(self, features) -> int:
        """Cluster index for a single vector (spark KMeansModel.predict)."""
        v = torch.as_tensor(
            features.toArray() if hasattr(features, "toArray") else features,
            dtype=torch.float64).reshape(1, -1)
        return int(_dist2(v, self._centers.to(v.device)).argmin())

    def clusterCenters(self):
        return [c.numpy() for c in self._centers]

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        X = dataset.col(self.getFeaturesCol()).to(torch.float64)
        C = self._centers.to(X.device)
        pred = _dist2(X, C).argmin(dim=1).to(torch.float64)
        return dataset.withColumn(self.getOrDefault(self.predictionCol), pred)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# GaussianMixture (spark:.../ml/clustering/GaussianMixture.scala)
# ---------------------------------------------------------------------------


class _GaussianMixtureParams(HasFeaturesCol, HasPredictionCol, HasProbabilityCol,
                             HasMaxIter, HasSeed, HasTol, HasWeightCol):
    k: Param[int] = Param(
        "undefined", "k", "Number of independent Gaussians in the mixture model.",
        TypeConverters.toInt)

    def __init__(self):
        super().__init__()
        self._setDefault(featuresCol="features", predictionCol="prediction",
                         probabilityCol="probability", k=2, maxIter=100,
                         tol=0.01, seed=42)


def _gmm_log_prob(X: torch.Tensor, mus: torch.Tensor, covs: torch.Tensor,
                  log_w: torch.Tensor) -> torch.Tensor:
    """log [w_k N(x | mu_k, Sigma_k)] per (row, component): N×k float64."""
    N, F = X.shape
    k = mus.shape[0]
    out = torch.empty(N, k, dtype=torch.float64, device=X.device)
    for j in range(k):
        L = torch.linalg.cholesky(covs[j])
        d = X - mus[j]
        z = torch.linalg.solve_triangular(L, d.T, upper=False)  # F×N
        maha = (z * z).sum(dim=0)
        logdet = 2.0 * torch.log(torch.diagonal(L)).sum()
        out[:, j] = log_w[j] - 0.5 * (
            maha + logdet + F * math.log(2.0 * math.pi))
    return out


class GaussianMixture(_GaussianMixtureParams, Estimator):
    """Full-covariance Gaussian mixture fit by EM
    (spark:.../ml/clustering/GaussianMixture.scala: defaults k=2,
    maxIter=100, tol=0.01 on the total log-likelihood delta).

    Replicated-state EM: every rank computes responsibilities over its
    shard, the sufficient statistics (N_k, Σ r x, Σ r x xᵀ, log-likelihood)
    are all-reduced in one flat buffer, and every rank applies the identical
    M-step (SURVEY.md §2.5 C2 pattern). A 1e-6 ridge keeps covariances SPD
    (documented divergence: Spark regularizes implicitly via its sample-cov
    init only).
    """

    def __init__(self, **kwargs):
        super().__init__()
        self._set(**kwargs)

    def setK(self, v: int):
        return self.set(self.k, v)

    def _fit(self, dataset: FlowFrame) -> "GaussianMixtureModel":
        X = dataset.col(self.getFeaturesCol())
        if not isinstance(X, torch.Tensor) or X.dim() != 2:
            raise TypeError("GaussianMixture features must be a vector column")
        Xd = X.to(torch.float64)
        N, F = Xd.shape
        k = self.getOrDefault(self.k)
        if k < 2:
            raise ValueError("k must be > 1")
        seed = self.getOrDefault(self.seed)
        g = torch.Generator(device=Xd.device if Xd.is_cuda else "cpu")
        g.manual_seed(seed & 0x7FFFFFFF)
        # global moments for the shared diagonal init covariance
        mom = torch.cat([
            torch.tensor([float(N)], dtype=torch.float64, device=Xd.device),
            Xd.sum(dim=0), (Xd * Xd).sum(dim=0)])
        dist.all_reduce_(mom)
        n_tot = float(mom[0])
        gmean = mom[1:1 + F] / n_tot
        gvar = (mom[1 + F:] / n_tot - gmean * gmean).clamp_min(1e-6)
        # init means from a short KMeans warm start (3 Lloyd iterations,
        # replicated/seeded — the same all-reduced path as KMeans.fit).
        # Documented divergence: Spark averages random sample slices, which
        # collapses to a saddle on symmetric mixtures; k-means init is
        # sklearn's default and strictly more robust.
        warm = KMeans(k=k, maxIter=3, seed=seed,
                      featuresCol=self.getFeaturesCol()).fit(dataset)
        mus = warm._centers.to(device=Xd.device, dtype=torch.float64).contiguous()
        dist.broadcast_(mus)
        covs = torch.diag_embed(gvar).expand(k, F, F).clone()
        w = torch.full((k,), 1.0 / k, dtype=torch.float64, device=Xd.device)
        ll_prev = None
        tol = self.getOrDefault(self.tol)
        it = 0
        ll = 0.0
        for it in range(self.getMaxIter()):
            lp = _gmm_log_prob(Xd, mus, covs, torch.log(w))
            lse = torch.logsumexp(lp, dim=1)
            r = torch.exp(lp - lse.unsqueeze(1))         # N×k
            # sufficient stats, one flat all_reduce
            Nk = r.sum(dim=0)                            # k
            Sx = r.T @ Xd                                # k×F
            Sxx = torch.stack([ (Xd * r[:, j:j + 1]).T @ Xd for j in range(k) ])
            buf = torch.cat([Nk, Sx.reshape(-1), Sxx.reshape(-1),
                             lse.sum().reshape(1)])
            dist.all_reduce_(buf)
            Nk = buf[:k]
            Sx = buf[k:k + k * F].view(k, F)
            Sxx = buf[k + k * F:k + k * F + k * F * F].view(k, F, F)
            ll = float(buf[-1])
            Nc = Nk.clamp_min(1e-12)
            w = Nk / n_tot
            mus = Sx / Nc.unsqueeze(1)
            covs = Sxx / Nc.view(k, 1, 1) - torch.einsum(
                "kf,kg->kfg", mus, mus)
            covs = covs + 1e-6 * torch.eye(F, dtype=torch.float64,
                                           device=Xd.device)
            if ll_prev is not None and abs(ll - ll_prev) < tol:
                ll_prev = ll
                break
            ll_prev = ll
        model = GaussianMixtureModel(weights=w.cpu(), mus=mus.cpu(),
                                     covs=covs.cpu(), logLikelihood=ll,
                                     numIter=it + 1)
        self._copyValues(model)
        model._set_parent(self)
        model._summary = ClusteringSummary(
            model, dataset, self.getOrDefault(self.predictionCol), k, it + 1,
            logLikelihood=ll,
            probabilityCol=self.getOrDefault(self.probabilityCol))
        return model


class GaussianMixtureModel(_GaussianMixtureParams, Model):
    def __init__(self, weights: Optional[torch.Tensor] = None,
                 mus: Optional[torch.Tensor] = None,
                 covs: Optional[torch.Tensor] = None,
                 logLikelihood: float = 0.0, numIter: int = 0):
        super().__init__()
        self._summary = None
        self._weights = weights if weights is not None else torch.zeros(0, dtype=torch.float64)
        self._mus = mus if mus is not None else torch.zeros(0, 0, dtype=torch.float64)
        self._covs = covs if covs is not None else torch.zeros(0, 0, 0, dtype=torch.float64)
        self.logLikelihood = float(logLikelihood)
        self.numIter = int(numIter)

    @property
    def summary(self) -> "ClusteringSummary":
        if self._summary is None:
            raise RuntimeError("No training summary available (absent after load)")
        return self._summary

    def hasSummary(self) -> bool:
        return self._summary is not None

    @property
    def weights(self) -> List[float]:
        return [float(v) for v in self._weights]

    @property
    def gaussians(self):
        """List of (mean DenseVector, cov DenseMatrix) like Spark's
        MultivariateGaussian pairs."""
        from flowml.linalg import DenseMatrix, DenseVector

        out = []
        for j in range(self._mus.shape[0]):
            mu = DenseVector(self._mus[j].numpy())
            F = self._covs.shape[1]
            cov = DenseMatrix(F, F, self._covs[j].T.reshape(-1).numpy())
            out.append((mu, cov))
        return out

    def predict(self, features) -> int:
        """Most-likely component for a single vector."""
        v = torch.as_tensor(
            features.toArray() if hasattr(features, "toArray") else features,
            dtype=torch.float32).reshape(1, -1)
        return int(self.predictProbability(v).argmax())

    def predictProbability(self, X: torch.Tensor) -> torch.Tensor:
        Xd = X.to(torch.float64)
        lp = _gmm_log_prob(Xd, self._mus.to(Xd.device), self._covs.to(Xd.device),
                           torch.log(self._weights.to(Xd.device)))
        return torch.exp(lp - torch.logsumexp(lp, dim=1, keepdim=True))

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        X = dataset.col(self.getFeaturesCol())
        prob = self.predictProbability(X)
        out = dataset.withColumn(
            self.getOrDefault(self.probabilityCol), prob.to(torch.float32))
        return out.withColumn(
            self.getOrDefault(self.predictionCol),
            prob.argmax(dim=1).to(torch.float64))

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


__all__ = ["KMeans", "KMeansModel", "BisectingKMeans", "BisectingKMeansModel",
           "GaussianMixture", "GaussianMixtureModel", "ClusteringSummary",
           "ClusteringEvaluator"]
