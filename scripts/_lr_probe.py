import time
import numpy as np
import torch
from flowml import ops

dev='cuda'
N, F, C = 1_250_000, 41, 5
r = np.random.default_rng(0)
X = torch.from_numpy(r.normal(size=(N, F)).astype(np.float32)).to(dev)
y = torch.from_numpy(r.integers(0, C, N).astype(np.int64)).to(dev)
W = torch.zeros(C, F, device=dev); b = torch.zeros(C, device=dev)
D = torch.from_numpy(r.normal(size=(N, C)).astype(np.float32)).to(dev)

def t(f, n=30):
    for _ in range(5): f()
    torch.cuda.synchronize(); t0=time.perf_counter()
    for _ in range(n): f()
    torch.cuda.synchronize(); return 1000*(time.perf_counter()-t0)/n

print("rocBLAS D.T@X     ", t(lambda: D.T @ X))
print("wgrad_f32 kernel  ", t(lambda: ops.get_ext().wgrad_f32(X, D)))
print("full logreg_grad  ", t(lambda: ops.logreg_grad(X, y, None, W, b)))
