"""Probe: LDS-atomic contention vs node spread in flowml_tree_hist_cls.

Same data, same launch shape; only the nodeid spread differs. If 1-node
(level-0-like) launches are much slower than 16-node launches, early levels
are contention-bound and wave-private LDS replication would pay.
"""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from flowml import ops

dev = "cuda"
N, F, B, C, T, Fp = 1_250_000, 41, 32, 5, 100, 6
r = np.random.default_rng(0)
codes = torch.from_numpy(r.integers(0, B, (N, F)).astype(np.uint8)).to(dev)
labels = torch.from_numpy(r.integers(0, C, N).astype(np.uint8)).to(dev)

def run(n_nodes, iters=20):
    G = T * n_nodes
    nid = torch.from_numpy(
        (r.integers(0, n_nodes, (T, N)) + np.arange(T)[:, None] * n_nodes).astype(np.int32)
    ).to(dev)
    lut = torch.arange(G, dtype=torch.int32, device=dev)
    tree_lo = torch.arange(0, G, n_nodes, dtype=torch.int32, device=dev)
    tree_cnt = torch.full((T,), n_nodes, dtype=torch.int32, device=dev)
    fs = torch.from_numpy(
        np.stack([np.sort(r.choice(F, Fp, replace=False)) for _ in range(G)]).astype(np.int32)
    ).to(dev)
    for _ in range(3):
        ops.tree_hist_nodes(codes, nid, lut, tree_lo, tree_cnt, labels, None, G, n_nodes, B, C, fs)
    torch.cuda.synchronize(); t = time.perf_counter()
    for _ in range(iters):
        ops.tree_hist_nodes(codes, nid, lut, tree_lo, tree_cnt, labels, None, G, n_nodes, B, C, fs)
    torch.cuda.synchronize()
    print(f"nodes/tree={n_nodes:3d}  {1000*(time.perf_counter()-t)/iters:.3f} ms")

for n in (1, 2, 4, 8, 16, 32):
    run(n)
