"""A/B: wgrad v2 (scalar transposed gathers) vs v3 (ds_read_b64_tr_b16).
Set FLOWML_WGRAD_V before launch (static env read)."""
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from flowml import ops

dev = "cuda"
ext = ops.get_ext()
print("FLOWML_WGRAD_V =", os.environ.get("FLOWML_WGRAD_V", "3 (default)"))
r = np.random.default_rng(0)
for (N, K, F) in ((1_000_000, 128, 41), (1_000_000, 64, 128), (1_000_000, 5, 64)):
    delta = torch.from_numpy(r.normal(size=(N, K)).astype(np.float32)).bfloat16().to(dev)
    act = torch.from_numpy(r.normal(size=(N, F)).astype(np.float32)).bfloat16().to(dev)
    for _ in range(5):
        ext.wgrad_bias(delta, act)
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(30):
        ext.wgrad_bias(delta, act)
    torch.cuda.synchronize()
    ms = 1000 * (time.perf_counter() - t0) / 30
    gw, gb = ext.wgrad_bias(delta, act)
    ref = (delta.double().T @ act.double()).float()
    err = (gw - ref).abs().max().item() / max(ref.abs().max().item(), 1e-9)
    print(f"K={K:4d} F={F:4d}: {ms:8.3f} ms  relerr={err:.2e}")
