#!/usr/bin/env python3
"""Unsupervised anomaly detection on flow records with GaussianMixture.

Fits a GMM to (mostly benign) traffic and flags the lowest-likelihood flows
as anomalies — the classic unsupervised complement to the supervised
classifiers in train_flows.py. Synthetic data: benign flows from two modes,
anomalies from a shifted distribution.

    python examples/detect_anomalies.py [--rows 20000] [--contamination 0.02]
    # data-parallel: python -m torch.distributed.run --nnodes=1 \
    #     --nproc-per-node 8 --master-addr 127.0.0.1 examples/detect_anomalies.py
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from flowml import FlowFrame, dist
from flowml.clustering import GaussianMixture
from flowml.feature import RobustScaler
from flowml.utils.config import RuntimeConfig


def synth(rows: int, contamination: float, seed: int):
    r = np.random.default_rng(seed)
    n_anom = int(rows * contamination)
    n_ben = rows - n_anom
    m1 = r.normal(loc=0.0, scale=1.0, size=(n_ben // 2, 8))
    m2 = r.normal(loc=[3, -2, 1, 0, 0, 2, -1, 0], scale=0.7,
                  size=(n_ben - n_ben // 2, 8))
    anom = r.normal(loc=8.0, scale=2.0, size=(n_anom, 8))
    X = np.vstack([m1, m2, anom]).astype(np.float32)
    y = np.concatenate([np.zeros(n_ben), np.ones(n_anom)])
    perm = r.permutation(rows)
    return X[perm], y[perm]


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=20_000)
    p.add_argument("--contamination", type=float, default=0.02)
    p.add_argument("--k", type=int, default=2)
    p.add_argument("--seed", type=int, default=7)
    p.add_argument("--device", default=None)
    args = p.parse_args()

    dist.init_from_env()
    cfg = RuntimeConfig()
    if args.device:
        cfg.device = args.device
    device = cfg.resolve_device()
    rank, world = dist.get_rank(), dist.get_world_size()

    X, y = synth(args.rows, args.contamination, args.seed)
    lo, hi = dist.shard_range(len(X))
    df = FlowFrame({"features": torch.from_numpy(X[lo:hi])}).to(device)

    scaler = RobustScaler(inputCol="features", outputCol="scaled",
                          withCentering=True).fit(df)
    df = scaler.transform(df)
    gmm = GaussianMixture(k=args.k, seed=args.seed,
                          featuresCol="scaled").fit(df)

    # anomaly score = negative max component log-likelihood
    Xs = df.col("scaled")
    from flowml.clustering import _gmm_log_prob

    lp = _gmm_log_prob(Xs.to(torch.float64),
                       gmm._mus.to(Xs.device), gmm._covs.to(Xs.device),
                       torch.log(gmm._weights.to(Xs.device)))
    score = -torch.logsumexp(lp, dim=1)
    # global threshold at the (1 - contamination) quantile of the score
    if world > 1:
        thr = float(dist.dist_quantiles(score, [1.0 - args.contamination])[0])
    else:
        thr = float(torch.quantile(score, 1.0 - args.contamination))
    flagged = (score > thr).cpu().numpy()
    y_local = y[lo:hi]
    tp = int(((flagged == 1) & (y_local == 1)).sum())
    fp = int(((flagged == 1) & (y_local == 0)).sum())
    fn = int(((flagged == 0) & (y_local == 1)).sum())
    agg = torch.tensor([tp, fp, fn], dtype=torch.float64)
    dist.all_reduce_(agg)
    tp, fp, fn = (int(v) for v in agg)
    prec = tp / max(tp + fp, 1)
    rec = tp / max(tp + fn, 1)
    if rank == 0:
        print(f"flows={args.rows} contamination={args.contamination} "
              f"k={args.k} gmm_iters={gmm.numIter}")
        print(f"anomaly precision={prec:.3f} recall={rec:.3f} "
              f"(threshold at the {1 - args.contamination:.0%} score quantile)")


if __name__ == "__main__":
    main()
