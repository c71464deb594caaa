"""Shared tree machinery: binning, level-synchronous histogram growth, ensembles.

Behavioral spec: ``spark:.../ml/tree/impl/RandomForest.scala`` + the ``tree/``
package: quantile bin thresholds from a row sample
(``findSplits``), uint8 bin codes (``TreePoint``), Poisson bagging
(``BaggedPoint``), per-node-group histogram aggregation (``DTStatsAggregator``)
and gini/variance split-gain selection (``binsToBestSplit``), under a
``maxMemoryInMB`` node-group budget.

MI355X-first redesign (SURVEY.md §1 control-flow invariant, §2.4 K2-K4):
growth is level-synchronous and *replicated* — every rank runs the identical
driver loop; only the histograms cross GPUs (one RCCL ``all_reduce`` per
node-group chunk, C2). One histogram pass serves ALL trees in the group
(inner loop over trees inside the K2 kernel), the binned uint8 matrix is the
only per-row data the hot loop reads (41 B/row), the split scan runs
redundantly on every rank (cheaper than a scatter), and all per-level driver
bookkeeping is numpy-vectorized — the host never loops over individual nodes
during growth.

Divergence from Spark, documented: all features are treated as continuous
(post-StringIndexer categorical indices are ordinal here; Spark's categorical
split-sets are not implemented — the saved-model ``split.numCategories`` is
written as 0/continuous).
"""

from __future__ import annotations

import math
import os
import time
from dataclasses import dataclass
from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from flowml import dist, ops


# ---------------------------------------------------------------------------
# Split-threshold discovery (reference: RandomForest.findSplits)
# ---------------------------------------------------------------------------


def find_split_thresholds(
    X: torch.Tensor, max_bins: int, seed: int, sample_cap: int = 1 << 16
) -> torch.Tensor:
    """Per-feature equi-depth quantile thresholds, F×(max_bins-1), +inf padded.

    Samples up to ``sample_cap`` rows (seeded) per rank, gathers samples
    across ranks (one-time, small — SURVEY.md §2.5 C6), sorts each feature
    column and takes lower quantiles, deduplicated. Fully torch-vectorized —
    runs on-device for the single-rank GPU path. All ranks produce identical
    thresholds. Bin semantics are strict: code = #(thr < x), so the binned
    split "code <= b" is exactly the raw-feature rule "x <= thr[b]" that
    inference (K8) applies — consistent even on integer-coded features.
    """
    N, F = X.shape
    if N > sample_cap:
        if X.is_cuda:
            # with-replacement sample on device (host randperm of the full
            # shard costs tens of ms); deterministic per (seed, device)
            g = torch.Generator(device=X.device).manual_seed(seed ^ 0x5EEDB1A5)
            idx = torch.randint(0, N, (sample_cap,), device=X.device, generator=g)
        else:
            g = torch.Generator().manual_seed(seed ^ 0x5EEDB1A5)
            idx = torch.randint(0, N, (sample_cap,), generator=g)
        sample = X.index_select(0, idx)
    else:
        sample = X
    if dist.is_initialized() and dist.get_world_size() > 1:
        parts = dist.all_gather_object(sample.detach().cpu().numpy())
        merged = np.concatenate(parts, axis=0)
        if len(merged) > 4 * sample_cap:
            rng = np.random.default_rng(seed ^ 0x5EEDB1A5)
            keep = rng.choice(len(merged), size=4 * sample_cap, replace=False)
            keep.sort()
            merged = merged[keep]
        sample = torch.from_numpy(np.ascontiguousarray(merged))
    B = max_bins
    S, _ = torch.sort(sample.to(torch.float32), dim=0)
    n = S.shape[0]
    qidx = (torch.arange(1, B, device=S.device, dtype=torch.int64) * (n - 1)) // B
    cand = S.index_select(0, qidx)  # (B-1)×F, ascending per column
    dup = torch.zeros_like(cand, dtype=torch.bool)
    dup[1:] = cand[1:] == cand[:-1]
    inf = torch.full_like(cand, float("inf"))
    cand = torch.where(dup, inf, cand)
    thr, _ = torch.sort(cand, dim=0)
    return thr.T.contiguous().to(X.device)


# ---------------------------------------------------------------------------
# Growth configuration + in-memory tree representation
# ---------------------------------------------------------------------------


@dataclass
class GrowthConfig:
    num_trees: int
    max_depth: int = 5
    max_bins: int = 32
    impurity: str = "gini"  # gini | entropy | variance
    feature_subset: str = "auto"  # auto|all|sqrt|onethird|log2|<int>|<float>
    min_instances_per_node: int = 1
    min_info_gain: float = 0.0
    max_memory_mb: int = 256
    subsampling_rate: float = 1.0
    bootstrap: bool = True
    seed: int = 0
    is_classification: bool = True
    n_classes: int = 2


class FlatTree:
    """Growing/grown tree in flat numpy arrays (node-local indices).

    Array-backed so the grower can append whole levels at once; public
    accessors return live slices (indexable and writable).
    """

    def __init__(self, stat_width: int = 0, cap: int = 8, regression: bool = False):
        self._n = 0
        self._sw = int(stat_width)
        self._cap = max(int(cap), 4)
        # Explicit impurity kind: a width-3 stats row is ambiguous between
        # 3-class classification counts and regression (s, s2, n), so the
        # kind must be threaded in, never shape-sniffed.
        self.regression = bool(regression)
        self._alloc(self._cap)

    def _alloc(self, cap: int) -> None:
        self._feature = np.full(cap, -1, np.int32)
        self._thr_bin = np.zeros(cap, np.int32)
        self._thr_raw = np.zeros(cap, np.float64)
        self._left = np.full(cap, -1, np.int32)
        self._right = np.full(cap, -1, np.int32)
        self._impurity = np.zeros(cap, np.float64)
        self._gain = np.zeros(cap, np.float64)
        self._count = np.zeros(cap, np.float64)
        self._stats = np.zeros((cap, self._sw), np.float64)

    def _ensure(self, extra: int) -> None:
        need = self._n + extra
        if need <= self._cap:
            return
        newcap = max(need, self._cap * 2)
        old = (self._feature, self._thr_bin, self._thr_raw, self._left,
               self._right, self._impurity, self._gain, self._count, self._stats)
        n = self._n
        self._cap = newcap
        self._alloc(newcap)
        for dst, src in zip(
            (self._feature, self._thr_bin, self._thr_raw, self._left,
             self._right, self._impurity, self._gain, self._count, self._stats),
            old,
        ):
            dst[:n] = src[:n]

    # -- live views ---------------------------------------------------------

    @property
    def n_nodes(self) -> int:
        return self._n

    @property
    def feature(self) -> np.ndarray:
        return self._feature[: self._n]

    @property
    def thr_bin(self) -> np.ndarray:
        return self._thr_bin[: self._n]

    @property
    def thr_raw(self) -> np.ndarray:
        return self._thr_raw[: self._n]

    @property
    def left(self) -> np.ndarray:
        return self._left[: self._n]

    @property
    def right(self) -> np.ndarray:
        return self._right[: self._n]

    @property
    def impurity(self) -> np.ndarray:
        return self._impurity[: self._n]

    @property
    def gain(self) -> np.ndarray:
        return self._gain[: self._n]

    @property
    def count(self) -> np.ndarray:
        return self._count[: self._n]

    @property
    def stats(self) -> np.ndarray:
        return self._stats[: self._n]

    # -- mutation -------------------------------------------------------------

    def add_node(self, stats, impurity: float, count: float) -> int:
        stats = np.asarray(stats, dtype=np.float64).reshape(-1)
        if self._sw == 0:
            self._sw = len(stats)
            self._stats = np.zeros((self._cap, self._sw), np.float64)
        self._ensure(1)
        i = self._n
        self._stats[i] = stats
        self._impurity[i] = impurity
        self._count[i] = count
        self._n += 1
        return i

    def bulk_add(self, stats2d: np.ndarray, imps: np.ndarray, cnts: np.ndarray) -> int:
        """Append k nodes; returns the base id (new nodes are base..base+k-1)."""
        k = len(imps)
        if self._sw == 0 and stats2d.size:
            self._sw = stats2d.shape[1]
            self._stats = np.zeros((self._cap, self._sw), np.float64)
        self._ensure(k)
        i = self._n
        self._stats[i : i + k] = stats2d
        self._impurity[i : i + k] = imps
        self._count[i : i + k] = cnts
        self._n += k
        return i

    def set_splits(self, nids, feats, bins, raws, lefts, rights, gains) -> None:
        self._feature[nids] = feats
        self._thr_bin[nids] = bins
        self._thr_raw[nids] = raws
        self._left[nids] = lefts
        self._right[nids] = rights
        self._gain[nids] = gains

    # -- queries --------------------------------------------------------------

    def prediction(self, i: int) -> float:
        s = self._stats[i]
        if self.regression:
            # stats = (sum, sum_sq, n)
            return float(s[0] / s[2]) if self._sw == 3 and s[2] > 0 else float(s[0])
        if self._sw > 1:
            return float(np.argmax(s))
        return float(s[0])

    def depth(self) -> int:
        if self._n == 0:
            return 0
        depth = np.zeros(self._n, np.int32)
        best = 0
        for i in range(self._n):
            if self._feature[i] >= 0:
                d = depth[i] + 1
                depth[self._left[i]] = d
                depth[self._right[i]] = d
                if d > best:
                    best = d
        return int(best)


def _feature_subset_size(strategy: str, F: int, is_classification: bool, num_trees: int) -> int:
    # Spark RandomForest.numFeaturesPerNode uses ceil for sqrt/log2/onethird
    # (e.g. F=41 → sqrt gives 7, not 6) — ADVICE r1 parity fix
    s = str(strategy)
    if s == "auto":
        if num_trees == 1:
            return F
        return (
            max(1, math.ceil(math.sqrt(F)))
            if is_classification
            else max(1, math.ceil(F / 3.0))
        )
    if s == "all":
        return F
    if s == "sqrt":
        return max(1, math.ceil(math.sqrt(F)))
    if s == "onethird":
        return max(1, math.ceil(F / 3.0))
    if s == "log2":
        return max(1, math.ceil(math.log2(F)))
    try:
        iv = int(s)
        if iv > 0:
            return min(F, iv)
    except ValueError:
        pass
    try:
        fv = float(s)
        if 0 < fv <= 1:
            return max(1, int(fv * F))
    except ValueError:
        pass
    raise ValueError(f"Unsupported featureSubsetStrategy: {strategy}")


def _impurity_from_stats(stats: torch.Tensor, impurity: str) -> torch.Tensor:
    """stats: ...×C class counts (classification) or ...×3 (s, s2, n)."""
    if impurity in ("gini", "entropy"):
        n = stats.sum(dim=-1).clamp_min(1e-12)
        p = stats / n.unsqueeze(-1)
        if impurity == "gini":
            return 1.0 - (p * p).sum(dim=-1)
        logp = torch.where(p > 0, torch.log2(p.clamp_min(1e-30)), torch.zeros_like(p))
        return -(p * logp).sum(dim=-1)
    # variance
    s, s2, n = stats[..., 0], stats[..., 1], stats[..., 2]
    nn = n.clamp_min(1e-12)
    return (s2 / nn - (s / nn) ** 2).clamp_min(0)


def _counts_of(stats: torch.Tensor, impurity: str) -> torch.Tensor:
    """Row count from a stats vector. MUST be keyed by the impurity kind:
    3-class classification stats also have width 3 — shape sniffing here
    silently read the class-2 count as n (bug found by the K3 kernel's
    explicit imp_code parity)."""
    if impurity == "variance":
        return stats[..., 2]
    return stats.sum(dim=-1)


def split_scan(
    hist: torch.Tensor,  # G×F'×B×S float32 (already all-reduced)
    impurity: str,
    min_instances: int,
    min_info_gain: float,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """Best split per node: returns (best_f', best_bin, best_gain, parent_stats,
    left_stats) — K3. Runs on whatever device hist lives on.
    """
    G, Fp, B, S = hist.shape
    h64 = hist.to(torch.float64)
    leftcum = h64.cumsum(dim=2)  # G×F'×B×S
    total = leftcum[:, :, B - 1, :]  # G×F'×S
    parent = total[:, 0, :]  # G×S  (same for all features)
    L = leftcum[:, :, : B - 1, :]  # candidate splits at bins 0..B-2
    R = total.unsqueeze(2) - L
    nL = _counts_of(L, impurity)
    nR = _counts_of(R, impurity)
    n = _counts_of(parent, impurity).clamp_min(1e-12)  # G
    impL = _impurity_from_stats(L, impurity)
    impR = _impurity_from_stats(R, impurity)
    imp_parent = _impurity_from_stats(parent, impurity)  # G
    child = (nL * impL + nR * impR) / n.view(G, 1, 1)
    gain = imp_parent.view(G, 1, 1) - child
    valid = (nL >= min_instances) & (nR >= min_instances)
    gain = torch.where(valid, gain, torch.full_like(gain, -float("inf")))
    flat = gain.reshape(G, -1)
    best = flat.argmax(dim=1)
    best_gain = flat.gather(1, best.unsqueeze(1)).squeeze(1)
    best_f = (best // (B - 1)).to(torch.int64)
    best_bin = (best % (B - 1)).to(torch.int64)
    ok = (best_gain > 0) & (best_gain >= min_info_gain) & torch.isfinite(best_gain)
    best_gain = torch.where(ok, best_gain, torch.zeros_like(best_gain))
    best_f = torch.where(ok, best_f, torch.full_like(best_f, -1))
    gidx = torch.arange(G, device=hist.device)
    left_stats = L[gidx, best_f.clamp_min(0), best_bin.clamp_min(0), :]
    return best_f, best_bin, best_gain, parent, left_stats


# ---------------------------------------------------------------------------
# The level-synchronous grower
# ---------------------------------------------------------------------------


def _hist_lds_cap(sub_F: int, B: int, S: int) -> int:
    """Max frontier nodes of ONE tree per K2 launch (per-tree LDS budget).

    The GPU kernel privatizes one tree's chunk histograms in LDS; 128 KiB of
    the 160 KiB/CU keeps room for staging. The torch path has no such limit
    but uses the same chunking so results are layout-identical.
    """
    bytes_per_node = sub_F * B * S * 4
    return max(1, (128 * 1024) // max(bytes_per_node, 1))


def _make_chunk_slices(fr_t: np.ndarray, T: int, g_max: int, l_cap: int) -> List[Tuple[int, int]]:
    """Contiguous [lo, hi) slices of the tree-major frontier obeying both the
    total-group budget (maxMemoryInMB) and the per-tree LDS cap. O(#trees)."""
    counts = np.bincount(fr_t, minlength=T)
    slices: List[Tuple[int, int]] = []
    lo = 0
    cur = 0
    pos = 0
    for t in range(T):
        c = int(counts[t])
        while c > 0:
            piece = min(c, l_cap, g_max)
            if cur and cur + piece > g_max:
                slices.append((lo, pos))
                lo = pos
                cur = 0
            cur += piece
            pos += piece
            c -= piece
            if c > 0:  # same tree continues: its next piece needs a new chunk
                slices.append((lo, pos))
                lo = pos
                cur = 0
    if pos > lo:
        slices.append((lo, pos))
    return slices


def grow_trees(
    codes: torch.Tensor,  # N×F uint8 (local shard)
    labels_or_targets: torch.Tensor,  # N int (classification) / N float (regression)
    bag: Optional[torch.Tensor],  # T×N weights or None (explicit tensor path)
    thresholds: torch.Tensor,  # F×(B-1) raw split values
    cfg: GrowthConfig,
    row_weights: Optional[torch.Tensor] = None,  # optional extra per-row weight
    hash_bag=None,  # (mode, cdf 8×f32, seed, global row offset): in-kernel bagging
    return_row_values: bool = False,  # reg only: also return per-(tree,row) leaf values
) -> List[FlatTree]:
    """Grow ``cfg.num_trees`` trees level-synchronously. Replicated across ranks:
    identical return on every rank (histograms are all-reduced).

    All nodes of all trees live in ONE struct-of-arrays arena (a FlatTree with
    global ids + a tree_of map): per-level bookkeeping is a handful of bulk
    numpy ops, the kernel's node->group lut is 1-D, and the K4b advance arrays
    are just the arena arrays — no per-tree host loops during growth.
    """
    N, F = codes.shape
    T = cfg.num_trees
    dev = codes.device
    B = cfg.max_bins
    is_cls = cfg.is_classification
    S = cfg.n_classes if is_cls else 3
    sub_F = _feature_subset_size(cfg.feature_subset, F, is_cls, T)
    use_subset = sub_F < F

    if bag is not None and row_weights is not None:
        bag = bag * row_weights.unsqueeze(0)
    elif row_weights is not None:
        bag = row_weights.unsqueeze(0).expand(T, N).contiguous()

    labels_u8 = labels_or_targets.to(torch.uint8) if is_cls else torch.zeros(N, dtype=torch.uint8, device=dev)
    targets = None if is_cls else labels_or_targets.to(torch.float32)

    timing = os.environ.get("FLOWML_TIMING", "0") == "1"
    tmr = {}

    def _tick():
        if timing and dev.type == "cuda":
            torch.cuda.synchronize()
        return time.perf_counter()

    def _acc(key, t0):
        if timing:
            tmr[key] = tmr.get(key, 0.0) + (_tick() - t0)

    hist_by_level = []  # per-level hist ms (timing only)

    arena = FlatTree(stat_width=S, cap=max(4 * T, 64), regression=not is_cls)
    tree_of_parts: List[np.ndarray] = [np.arange(T, dtype=np.int64)]
    # roots are arena nodes 0..T-1; every (tree,row) starts at its root
    nodeid = (
        torch.arange(T, dtype=torch.int32, device=dev).unsqueeze(1).expand(T, N).contiguous()
    )

    min_c = 2 * cfg.min_instances_per_node
    if cfg.max_depth == 0:
        # root stats per tree: one K2 pass over feature 0 (lut: root t -> t)
        lut0 = torch.arange(T, dtype=torch.int32, device=dev)
        tree_lo0 = torch.arange(T, dtype=torch.int32, device=dev)
        tree_cnt0 = torch.ones(T, dtype=torch.int32, device=dev)
        fs0 = torch.zeros((T, 1), dtype=torch.int32, device=dev)
        h0 = ops.tree_hist_nodes(
            codes, nodeid, lut0, tree_lo0, tree_cnt0, labels_u8, bag, T, 1, B,
            S if is_cls else 0, fs0, targets=targets, hash_bag=hash_bag,
        )
        dist.all_reduce_(h0.reshape(-1))
        if is_cls:
            root_stats = h0.sum(dim=2)[:, 0, :].to(torch.float64)  # T×S
            root_imp = _impurity_from_stats(root_stats, cfg.impurity)
            root_cnt = _counts_of(root_stats, cfg.impurity)
        else:
            cells0, ns20 = ops.reg_hist_views(h0, T, 1, B)
            s0 = cells0[:, 0, :, 0].to(torch.float64).sum(-1)
            n0 = cells0[:, 0, :, 1].to(torch.float64).sum(-1)
            s20 = ns20.to(torch.float64)
            n0c = n0.clamp_min(1e-12)
            m0 = s0 / n0c
            root_imp = (s20 / n0c - m0 * m0).clamp_min(0.0)
            root_cnt = n0
            root_stats = torch.stack([s0, s20, n0], dim=1)  # T×3 (s, s², n)
        rs_np = root_stats.cpu().numpy()
        ri_np = root_imp.cpu().numpy()
        rc_np = root_cnt.cpu().numpy()
        arena.bulk_add(rs_np, ri_np, rc_np)
        grow_ok = (rc_np >= min_c) & (ri_np > 0) & (cfg.max_depth > 0)
        fr_t = np.nonzero(grow_ok)[0].astype(np.int64)  # frontier tree ids
        fr_n = fr_t.copy()  # global arena ids of the roots
    else:
        # the dedicated root pass is redundant when the trees grow: the
        # level-0 scan emits each root's own stats/impurity from its
        # histogram (trailing packed columns), and its min_instances check
        # turns too-small/pure roots into leaves. Saves one K2 pass per fit
        # (one per ROUND for GBT).
        arena.bulk_add(np.zeros((T, S)), np.full(T, np.nan), np.zeros(T))
        fr_t = np.arange(T, dtype=np.int64)
        fr_n = fr_t.copy()

    # memory budget → groups per histogram chunk (maxMemoryInMB parity) and
    # per-tree LDS cap (GPU kernel constraint)
    S_hist = S if is_cls else 2  # reg cells are (Σwt, Σw); Σt² is per-node
    bytes_per_node = sub_F * B * S_hist * 4
    g_max = max(1, int(cfg.max_memory_mb * 1024 * 1024 // max(bytes_per_node, 1)))
    l_cap = _hist_lds_cap(sub_F, B, S_hist)

    thr_host = thresholds.cpu().numpy()
    level = 0
    pending_adv = None  # fused K4b: applied inside the first K2 chunk of a level
    hist_global_env = int(os.environ.get("FLOWML_HIST_GLOBAL", "-1"))
    while len(fr_t) and level < cfg.max_depth:
        nxt_t: List[np.ndarray] = []
        nxt_n: List[np.ndarray] = []
        # Deep levels overflow the per-tree LDS budget and would force one
        # full pass over the N×F codes PER LDS-sized chunk. The kernel can
        # instead accumulate with fp32 HBM atomics in ONE pass (l_max=0 →
        # GLOBAL mode); only the maxMemoryInMB budget still chunks.
        # Classification counts are integers ⇒ exact/bitwise in any order;
        # regression sums were already order-dependent at the LDS flush, so
        # GLOBAL mode is the same accuracy class. Decision is
        # device-independent (counts only) so the torch oracle sees
        # identical chunking → parity holds.
        slices = _make_chunk_slices(fr_t, T, g_max, l_cap)
        use_global = False
        if hist_global_env != 0 and len(slices) > 2:
            budget_slices = _make_chunk_slices(fr_t, T, g_max, g_max)
            if hist_global_env == 1 or len(slices) >= 2 * len(budget_slices):
                slices = budget_slices
                use_global = True
        for ci, (clo, chi) in enumerate(slices):
            t_chunk0 = _tick()
            chunk_t = fr_t[clo:chi]
            chunk_n = fr_n[clo:chi]
            G = chi - clo
            # one packed [lut | tree_lo | tree_cnt] array → ONE H2D copy
            # (three small transfers per chunk are latency-bound on PCIe)
            n_nodes = arena.n_nodes
            packed_np = np.empty(n_nodes + 2 * T, dtype=np.int32)
            packed_np[:n_nodes] = -1
            packed_np[chunk_n] = np.arange(G, dtype=np.int32)
            per_tree_count = np.bincount(chunk_t, minlength=T).astype(np.int32)
            tree_lo_np = packed_np[n_nodes : n_nodes + T]
            tree_lo_np[:] = 0
            nz = per_tree_count > 0
            starts = np.cumsum(per_tree_count) - per_tree_count
            tree_lo_np[nz] = starts[nz]
            packed_np[n_nodes + T :] = per_tree_count
            l_max = int(per_tree_count.max())
            packed_d = torch.from_numpy(packed_np).to(dev)
            lut_d = packed_d[:n_nodes]
            tree_lo = packed_d[n_nodes : n_nodes + T]
            tree_cnt = packed_d[n_nodes + T :]
            # per-node feature subsets: one seeded RNG per (level, chunk) —
            # the frontier order is deterministic and identical on every rank,
            # so this is rank-invariant and reproducible (√F draw, K2 input)
            if use_subset:
                r = np.random.default_rng((cfg.seed * 1_000_003 + level * 8191 + ci) & 0x7FFFFFFF)
                # vectorized partial Fisher-Yates: sub_F int draws per node
                # instead of F random keys + argpartition (the fs draw was
                # the largest host-side cost of the flagship fit)
                pool = np.broadcast_to(
                    np.arange(F, dtype=np.int32), (G, F)
                ).copy()
                rows_ar = np.arange(G)
                for j in range(sub_F):
                    k = r.integers(j, F, G)
                    tmp = pool[rows_ar, j].copy()
                    pool[rows_ar, j] = pool[rows_ar, k]
                    pool[rows_ar, k] = tmp
                fs_np = np.sort(pool[:, :sub_F], axis=1)
                fs = torch.from_numpy(fs_np).to(dev)
            else:
                fs_np = None
                fs = None
            _acc("chunk_setup", t_chunk0)
            t0 = _tick()
            hist = ops.tree_hist_nodes(
                codes, nodeid, lut_d, tree_lo, tree_cnt, labels_u8, bag, G,
                0 if use_global else l_max, B,
                S if is_cls else 0, fs, targets=targets,
                advance=pending_adv if ci == 0 else None, hash_bag=hash_bag,
            )
            if ci == 0:
                pending_adv = None
            if timing:
                while len(hist_by_level) <= level:
                    hist_by_level.append(0.0)
                hist_by_level[level] += (_tick() - t0) * 1000
            _acc("hist", t0)
            t0 = _tick()
            dist.all_reduce_(hist.reshape(-1))
            _acc("allreduce", t0)
            t0 = _tick()
            # K3 split-gain scan: packed per-node results, one host transfer
            if is_cls:
                packed = ops.split_scan_packed(
                    hist, cfg.impurity, cfg.min_instances_per_node, cfg.min_info_gain
                ).cpu().numpy().astype(np.float64)
            else:
                packed = ops.split_scan_packed_reg(
                    hist, G, sub_F, B, cfg.min_instances_per_node, cfg.min_info_gain
                ).cpu().numpy().astype(np.float64)
            _acc("scan", t0)
            t0 = _tick()
            bf = packed[:, 0].astype(np.int64)
            bb = packed[:, 1].astype(np.int64)
            bg = packed[:, 2]
            if is_cls:
                il = packed[:, 3]
                ir = packed[:, 4]
                cl = packed[:, 5]
                cr = packed[:, 6]
                ls = packed[:, 7 : 7 + S]
                rs = packed[:, 7 + S : 7 + 2 * S]
                # node-own impurity/stats (trailing columns): covers roots,
                # and is a bitwise no-op for nodes created by a parent scan
                arena.impurity[chunk_n] = packed[:, 7 + 2 * S]
                pstats = packed[:, 8 + 2 * S : 8 + 3 * S]
                arena.stats[chunk_n] = pstats
                arena.count[chunk_n] = pstats.sum(axis=1)
            else:
                # [f, b, gain, sL, nL, sR, nR, imp_self, s_self, n_self]:
                # child impurities are unknown here (backfilled after growth)
                cl = packed[:, 4]
                cr = packed[:, 6]
                il = np.full(len(packed), np.nan)
                ir = np.full(len(packed), np.nan)
                ls = np.zeros((len(packed), S))
                ls[:, 0] = packed[:, 3]
                ls[:, 2] = packed[:, 4]
                rs = np.zeros((len(packed), S))
                rs[:, 0] = packed[:, 5]
                rs[:, 2] = packed[:, 6]
                # this node's own impurity/stats are now known exactly:
                # record (also covers roots — no dedicated root pass for reg)
                arena.impurity[chunk_n] = packed[:, 7]
                arena.stats[chunk_n, 0] = packed[:, 8]
                arena.stats[chunk_n, 2] = packed[:, 9]
                arena.count[chunk_n] = packed[:, 9]
                arena.stats[chunk_n, 1] = np.asarray(
                    ops.reg_hist_views(hist, G, sub_F, B)[1].cpu().numpy(),
                    dtype=np.float64)
            sel = np.nonzero(bf >= 0)[0]
            if len(sel):
                k = len(sel)
                gf = fs_np[sel, bf[sel]] if fs_np is not None else bf[sel]
                raws = thr_host[gf, bb[sel]]
                # bulk-append ALL children of the chunk to the arena at once
                stats2d = np.empty((2 * k, S))
                stats2d[0::2] = ls[sel]
                stats2d[1::2] = rs[sel]
                imps = np.empty(2 * k)
                imps[0::2] = il[sel]
                imps[1::2] = ir[sel]
                cnts = np.empty(2 * k)
                cnts[0::2] = cl[sel]
                cnts[1::2] = cr[sel]
                base = arena.bulk_add(stats2d, imps, cnts)
                lid = base + 2 * np.arange(k, dtype=np.int64)
                rid = lid + 1
                arena.set_splits(chunk_n[sel], gf, bb[sel], raws, lid, rid, bg[sel])
                tree_of_parts.append(np.repeat(chunk_t[sel], 2))
                if level + 1 < cfg.max_depth:
                    if is_cls:
                        keep_l = (cl[sel] >= min_c) & (il[sel] > 0)
                        keep_r = (cr[sel] >= min_c) & (ir[sel] > 0)
                    else:
                        # child impurity is unknown until its own scan; a
                        # pure child resolves next level via gain == 0
                        keep_l = cl[sel] >= min_c
                        keep_r = cr[sel] >= min_c
                    mask = np.ravel(np.column_stack([keep_l, keep_r]))
                    nxt_t.append(np.repeat(chunk_t[sel], 2)[mask])
                    nxt_n.append(np.column_stack([lid, rid]).ravel()[mask])
            _acc("bookkeep", t0)
        if nxt_t:
            fr_t = np.concatenate(nxt_t)
            fr_n = np.concatenate(nxt_n)
        else:
            fr_t = np.zeros(0, np.int64)
            fr_n = np.zeros(0, np.int64)
        if not len(fr_t):
            break
        # stage the K4b advance arrays (= the arena arrays) for the next
        # level's first K2 chunk — fused: one pass over the shard, not two
        t0 = _tick()
        pending_adv = (
            torch.from_numpy(arena.feature.copy()).to(dev),
            torch.from_numpy(arena.thr_bin.copy()).to(dev),
            torch.from_numpy(arena.left.copy()).to(dev),
            torch.from_numpy(arena.right.copy()).to(dev),
        )
        _acc("advance", t0)
        level += 1

    tree_of = np.concatenate(tree_of_parts)

    if not is_cls and arena.n_nodes > T:
        # rows lag one level behind the arena (the fused advance runs at the
        # START of the next level's hist, so children created by the FINAL
        # scan never receive their rows): one standalone advance settles
        # every row at its true leaf (idempotent at leaves — keep-at-leaf).
        t0 = _tick()
        nodeid.copy_(ops.node_advance(
            codes, nodeid,
            torch.from_numpy(arena.feature.copy()).to(dev),
            torch.from_numpy(arena.thr_bin.copy()).to(dev),
            torch.from_numpy(arena.left.copy()).to(dev),
            torch.from_numpy(arena.right.copy()).to(dev),
        ))
        _acc("final_advance", t0)

    if not is_cls:
        # backfill leaf Σt²/impurity: rows sit at their leaves (keep-at-leaf
        # advance), so ONE single-feature K2b pass over the never-scanned
        # nodes yields their per-node Σwt² slots. (A torch index_add was
        # 1.27 ms/round here — 6M rows funneling into ~60 leaf slots
        # serializes atomics; the hist kernel privatizes per-block in LDS.)
        # Values are informational (NodeData impurity / impurityStats) —
        # split decisions never read them.
        t0 = _tick()
        need = np.isnan(arena.impurity)
        if need.any():
            need_idx = np.nonzero(need)[0]
            grp = need_idx[np.argsort(tree_of[need_idx], kind="stable")]
            Gn = len(grp)
            lut_np2 = np.full(arena.n_nodes, -1, dtype=np.int32)
            lut_np2[grp] = np.arange(Gn, dtype=np.int32)
            cnt_np2 = np.bincount(tree_of[grp], minlength=T).astype(np.int32)
            lo_np2 = (np.cumsum(cnt_np2) - cnt_np2).astype(np.int32)
            l_max_n = int(cnt_np2.max())
            if l_max_n * (B * 2 + 1) * 4 > 128 * 1024:
                l_max_n = 0  # deep forest: GLOBAL mode
            hb = ops.tree_hist_nodes(
                codes, nodeid,
                torch.from_numpy(lut_np2).to(dev),
                torch.from_numpy(lo_np2).to(dev),
                torch.from_numpy(cnt_np2).to(dev),
                labels_u8, bag, Gn, l_max_n, B, 0,
                torch.zeros((Gn, 1), dtype=torch.int32, device=dev),
                targets=targets, hash_bag=hash_bag,
            )
            dist.all_reduce_(hb.reshape(-1))
            ns2b = ops.reg_hist_views(hb, Gn, 1, B)[1].cpu().numpy().astype(np.float64)
            stats = arena.stats
            stats[grp, 1] = ns2b
            n_np = np.maximum(stats[:, 2], 1e-12)
            m_np = stats[:, 0] / n_np
            imp_np = np.maximum(stats[:, 1] / n_np - m_np * m_np, 0.0)
            arena.impurity[need] = imp_np[need]
        _acc("leaf_backfill", t0)

    row_values = None
    if return_row_values and not is_cls:
        # per-(tree,row) leaf predictions by a device gather over the settled
        # nodeid — bit-identical to traversing the built tree (strict-< bin
        # consistency), ~50 µs vs a 0.5 ms K8 re-predict (GBT margin update)
        vals_np = (arena.stats[:, 0] /
                   np.maximum(arena.count, 1e-300)).astype(np.float32)
        vals_np[arena.count <= 0] = 0.0
        node_vals = torch.from_numpy(vals_np).to(dev)
        row_values = node_vals[nodeid.to(torch.int64)]

    # extract per-tree FlatTrees from the arena (once per fit, vectorized)
    t0 = _tick()
    order = np.argsort(tree_of, kind="stable")  # per-tree contiguous, BFS order
    bounds = np.searchsorted(tree_of[order], np.arange(T + 1))
    g2l = np.empty(arena.n_nodes, dtype=np.int64)
    g2l[order] = np.concatenate(
        [np.arange(bounds[t + 1] - bounds[t]) for t in range(T)]
    ) if T else np.zeros(0, np.int64)
    # gather the whole arena into per-tree-contiguous order ONCE, then each
    # tree is a cheap contiguous-slice copy (one gather beats T fancy-index
    # passes — this was 1.5 ms of the 20 ms flagship fit)
    st_o = arena.stats[order]
    imp_o = arena.impurity[order]
    cnt_o = arena.count[order]
    feat_o = arena.feature[order]
    thb_o = arena.thr_bin[order]
    thr_o = arena.thr_raw[order]
    gain_o = arena.gain[order]
    internal_o = feat_o >= 0
    left_o = np.where(internal_o, g2l[np.where(internal_o, arena.left[order], 0)], -1)
    right_o = np.where(internal_o, g2l[np.where(internal_o, arena.right[order], 0)], -1)
    trees: List[FlatTree] = []
    for t in range(T):
        sl = slice(bounds[t], bounds[t + 1])
        n = bounds[t + 1] - bounds[t]
        tr = FlatTree(stat_width=S, cap=max(int(n), 4), regression=not is_cls)
        tr.bulk_add(st_o[sl], imp_o[sl], cnt_o[sl])
        tr.feature[:] = feat_o[sl]
        tr.thr_bin[:] = thb_o[sl]
        tr.thr_raw[:] = thr_o[sl]
        tr.gain[:] = gain_o[sl]
        tr.left[:] = left_o[sl]
        tr.right[:] = right_o[sl]
        trees.append(tr)
    _acc("extract", t0)

    if timing:
        print("grow_trees timing:", {k: round(v * 1000, 1) for k, v in tmr.items()})
        print("hist ms by level:", [round(v, 2) for v in hist_by_level])
    if return_row_values:
        return trees, row_values
    return trees


# ---------------------------------------------------------------------------
# Ensemble container → K8 inference arrays + importances
# ---------------------------------------------------------------------------


class TreeEnsemble:
    """Inference-ready ensemble: padded node arrays for the K8 kernel."""

    def __init__(self, trees: List[FlatTree], n_classes: int, mode: str = "prob_sum"):
        self.trees = trees
        self.n_classes = n_classes
        self.mode = mode  # prob_sum (RF) | scalar (GBT margins)
        T = len(trees)
        M = max((t.n_nodes for t in trees), default=1)
        K = n_classes if mode == "prob_sum" else 1
        feature = np.full((T, M), -1, dtype=np.int32)
        threshold = np.zeros((T, M), dtype=np.float32)
        left = np.zeros((T, M), dtype=np.int32)
        right = np.zeros((T, M), dtype=np.int32)
        value = np.zeros((T, M, K), dtype=np.float32)
        for ti, tr in enumerate(trees):
            n = tr.n_nodes
            if n == 0:
                continue
            feature[ti, :n] = tr.feature
            threshold[ti, :n] = tr.thr_raw
            left[ti, :n] = tr.left
            right[ti, :n] = tr.right
            stats = tr.stats
            if mode == "prob_sum":
                tot = stats.sum(axis=1, keepdims=True)
                value[ti, :n] = np.divide(
                    stats, tot, out=np.zeros_like(stats), where=tot > 0
                )
            else:
                cnt = stats[:, 2]
                value[ti, :n, 0] = np.divide(
                    stats[:, 0], cnt, out=np.zeros(n), where=cnt > 0
                )
        self._feature = torch.from_numpy(feature)
        self._threshold = torch.from_numpy(threshold)
        self._left = torch.from_numpy(left)
        self._right = torch.from_numpy(right)
        self._value = torch.from_numpy(value)
        self._tree_weights: Optional[torch.Tensor] = None  # GBT stepSize weights

    def set_tree_weights(self, w: Sequence[float]) -> None:
        self._tree_weights = torch.tensor(list(w), dtype=torch.float32)

    def to(self, device) -> "TreeEnsemble":
        self._feature = self._feature.to(device)
        self._threshold = self._threshold.to(device)
        self._left = self._left.to(device)
        self._right = self._right.to(device)
        self._value = self._value.to(device)
        if self._tree_weights is not None:
            self._tree_weights = self._tree_weights.to(device)
        return self

    def predict_raw(self, X: torch.Tensor) -> torch.Tensor:
        """N×K sum of per-tree leaf vectors (weighted for GBT)."""
        self.to(X.device)
        value = self._value
        if self._tree_weights is not None:
            value = (value * self._tree_weights.view(-1, 1, 1)).contiguous()
        return ops.tree_predict(
            X.to(torch.float32), self._feature, self._threshold, self._left, self._right, value
        )

    def feature_importances(self, num_features: int) -> np.ndarray:
        """Spark semantics: per tree, Σ_internal gain·count, normalized per
        tree, summed, normalized again (spark: TreeEnsembleModel.featureImportances)."""
        total = np.zeros(num_features, dtype=np.float64)
        for tr in self.trees:
            internal = tr.feature >= 0
            contrib = np.zeros(num_features, dtype=np.float64)
            np.add.at(contrib, tr.feature[internal], tr.gain[internal] * tr.count[internal])
            s = contrib.sum()
            if s > 0:
                total += contrib / s
        s = total.sum()
        return total / s if s > 0 else total


def poisson_bag(
    n_rows: int, n_trees: int, rate: float, bootstrap: bool, seed: int, device
) -> Optional[torch.Tensor]:
    """Per-(tree,row) Poisson bag counts, shape (T, N) (BaggedPoint; §3.2).

    Seeded per rank by (seed, rank) — deterministic for a fixed world size.
    Returns None when no subsampling is requested (weight ≡ 1).
    """
    if not bootstrap and rate >= 1.0:
        return None
    device = torch.device(device)
    seed_r = (seed * 7919 + dist.get_rank()) & 0x7FFFFFFF
    if device.type == "cuda":
        # generate on-device (a 100×10M Poisson draw on host would dominate);
        # deterministic per (seed, rank) via a dedicated CUDA generator
        g = torch.Generator(device=device).manual_seed(seed_r)
        if bootstrap:
            bag = torch.poisson(torch.full((n_trees, n_rows), rate, device=device), generator=g)
        else:
            bag = (torch.rand((n_trees, n_rows), device=device, generator=g) < rate).to(torch.float32)
        return bag.to(torch.float32)
    g = torch.Generator(device="cpu").manual_seed(seed_r)
    if bootstrap:
        bag = torch.poisson(torch.full((n_trees, n_rows), rate), generator=g)
    else:
        bag = (torch.rand((n_trees, n_rows), generator=g) < rate).to(torch.float32)
    return bag.to(torch.float32).to(device)
