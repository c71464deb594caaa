"""Distributed runtime: RCCL over xGMI via torch.distributed.

Replaces the Spark core runtime: driver/executor RPC,
``RDD.treeAggregate`` and ``TorrentBroadcast`` collapse to one process per
GPU with RCCL collectives (``torch.distributed`` backend "nccl" IS RCCL on
ROCm). Design rules from SURVEY.md §2.5/§5.8:

* the replicated-state pattern: after every ``all_reduce(sum)`` each rank
  applies the identical deterministic update, so per-iteration broadcast is
  elided (C4);
* the many small per-step reduces are fused into one flat fp32 buffer
  (``FusedAllReduce``) — payloads are latency-bound (≤8 MB) on the
  fully-connected 7×153 GB/s xGMI mesh, so fewer, larger collectives win;
* gloo on CPU is the test backend (§4 "multi-node-without-a-cluster"); when
  no process group is initialized every collective is the identity
  (world_size=1 semantics), so all single-process code runs unchanged.
"""

from __future__ import annotations

import datetime
import math
import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as td


def is_initialized() -> bool:
    return td.is_available() and td.is_initialized()


def get_rank() -> int:
    return td.get_rank() if is_initialized() else 0


def get_world_size() -> int:
    return td.get_world_size() if is_initialized() else 1


def init_from_env(backend: Optional[str] = None, timeout_s: float = 600.0) -> None:
    """Initialize the process group from torchrun env vars (env:// rendezvous).

    Picks RCCL ("nccl") when a GPU is visible, else gloo. Idempotent.
    """
    if is_initialized():
        return
    if "RANK" not in os.environ:
        return  # single-process mode
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if backend == "nccl":
        local_rank = int(os.environ.get("LOCAL_RANK", "0"))
        torch.cuda.set_device(local_rank)
    td.init_process_group(backend=backend, timeout=datetime.timedelta(seconds=timeout_s))


def all_reduce_(tensor: torch.Tensor, op: str = "sum") -> torch.Tensor:
    """In-place reduce across ranks (identity when single-process).

    op ∈ {"sum", "max", "min"} — both RCCL and gloo support all three.
    """
    if is_initialized():
        ops = {"sum": td.ReduceOp.SUM, "max": td.ReduceOp.MAX, "min": td.ReduceOp.MIN}
        td.all_reduce(tensor, op=ops[op])
    return tensor


def broadcast_(tensor: torch.Tensor, src: int = 0) -> torch.Tensor:
    if is_initialized():
        td.broadcast(tensor, src=src)
    return tensor


def barrier() -> None:
    if is_initialized():
        td.barrier()


def all_gather_object(obj) -> List:
    if not is_initialized():
        return [obj]
    out = [None] * get_world_size()
    td.all_gather_object(out, obj)
    return out


def shard_range(n: int, rank: Optional[int] = None, world: Optional[int] = None):
    """Contiguous static row shard [lo, hi) for this rank."""
    r = get_rank() if rank is None else rank
    w = get_world_size() if world is None else world
    lo = (n * r) // w
    hi = (n * (r + 1)) // w
    return lo, hi


def dist_quantiles(x: torch.Tensor, probs: Sequence[float], iters: int = 64) -> torch.Tensor:
    """Global type-1 quantiles of 1-D data sharded across ranks.

    Replaces Spark's ``approxQuantile`` (GK sketch, driver merge) for the
    feature estimators: bisection on the value range with a count all-reduce
    per iteration (≈65 tiny collectives), then an exact min-reduce to land on
    a real data value. Deterministic and rank-count-invariant: every rank
    returns the identical tensor. Single-process calls never reach here —
    callers keep torch.quantile's interpolated local path for world==1.
    """
    x = x.reshape(-1).to(torch.float64)
    P = len(probs)
    dev = x.device
    n_t = torch.tensor([float(x.numel())], dtype=torch.float64, device=dev)
    all_reduce_(n_t)
    n = float(n_t.item())
    if n == 0:
        return torch.zeros(P, dtype=torch.float64, device=dev)
    lo_t = (x.min() if x.numel() else torch.tensor(float("inf"), dtype=torch.float64, device=dev)).reshape(1).clone()
    hi_t = (x.max() if x.numel() else torch.tensor(float("-inf"), dtype=torch.float64, device=dev)).reshape(1).clone()
    all_reduce_(lo_t, op="min")
    all_reduce_(hi_t, op="max")
    # target rank k_p = ceil(p·n) clamped to [1, n]; answer = smallest value v
    # with global count(x <= v) >= k_p
    ks = torch.tensor(
        [min(max(math.ceil(p * n), 1), int(n)) for p in probs], dtype=torch.float64, device=dev
    )
    # invariant: count(x <= lo) < k <= count(x <= hi); start lo strictly
    # below the global min so the invariant holds for k = 1
    span = (hi_t - lo_t).clamp_min(1.0)
    lo = (lo_t - span).expand(P).contiguous().clone()
    hi = hi_t.expand(P).contiguous().clone()
    for _ in range(iters):
        mid = (lo + hi) / 2
        cnt = (x.unsqueeze(0) <= mid.unsqueeze(1)).sum(dim=1).to(torch.float64)
        all_reduce_(cnt)
        ge = cnt >= ks
        hi = torch.where(ge, mid, hi)
        lo = torch.where(ge, lo, mid)
    # exact landing: smallest data value in (lo, hi]
    big = torch.full((P,), float("inf"), dtype=torch.float64, device=dev)
    if x.numel():
        masked = torch.where(
            (x.unsqueeze(0) > lo.unsqueeze(1)) & (x.unsqueeze(0) <= hi.unsqueeze(1)),
            x.unsqueeze(0),
            big.unsqueeze(1),
        )
        cand = masked.min(dim=1).values
    else:
        cand = big.clone()
    all_reduce_(cand, op="min")
    return torch.where(torch.isinf(cand), hi, cand)


class FusedAllReduce:
    """Fuse several small tensors into one flat buffer per all_reduce (C1-C3).

    Usage: register tensors once, then ``reduce()`` each iteration — it packs,
    all-reduces the single flat buffer, and unpacks in place. The flat buffer
    is persistent so the collective is hipGraph-capturable.
    """

    def __init__(self, tensors: Sequence[torch.Tensor], dtype: torch.dtype = torch.float32):
        self.tensors = list(tensors)
        self.numel = sum(t.numel() for t in self.tensors)
        dev = self.tensors[0].device if self.tensors else torch.device("cpu")
        self.flat = torch.zeros(self.numel, dtype=dtype, device=dev)
        self.views = []
        off = 0
        for t in self.tensors:
            n = t.numel()
            self.views.append(self.flat[off : off + n].view(t.shape))
            off += n

    def reduce(self, force: bool = False) -> None:
        """Pack → all_reduce(flat) → unpack.

        At world==1 the collective is skipped unless ``force`` (or env
        ``FLOWML_FORCE_COLLECTIVES=1``) — forcing lets a single-GPU run
        exercise the live RCCL communicator on the real payload (including
        under hipGraph capture), so N>1 behavior is de-risked before an
        8-GPU lease exists.
        """
        if not is_initialized():
            return
        if get_world_size() == 1 and not force and os.environ.get(
            "FLOWML_FORCE_COLLECTIVES", "0"
        ) != "1":
            return
        for t, v in zip(self.tensors, self.views):
            v.copy_(t)
        all_reduce_(self.flat)
        for t, v in zip(self.tensors, self.views):
            t.copy_(v)
