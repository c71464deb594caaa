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
only per-row data the hot loop reads (41 B/row), and the split scan runs
redundantly on every rank (cheaper than a scatter).

Divergence from Spark, documented: all features are treated as continuous
(post-StringIndexer categorical indices are ordinal here; Spark's categorical
split-sets are not implemented — the saved-model ``split.numCategories`` is
written as 0/continuous).
"""

from __future__ import annotations

import math
import os
import time
from dataclasses import dataclass, field
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


@dataclass
class FlatTree:
    """Growing/grown tree in flat arrays (node-local indices)."""

    feature: List[int] = field(default_factory=list)  # -1 = leaf
    thr_bin: List[int] = field(default_factory=list)
    thr_raw: List[float] = field(default_factory=list)
    left: List[int] = field(default_factory=list)
    right: List[int] = field(default_factory=list)
    stats: List[np.ndarray] = field(default_factory=list)  # class counts / (s,s2,n)
    impurity: List[float] = field(default_factory=list)
    gain: List[float] = field(default_factory=list)
    count: List[float] = field(default_factory=list)

    def add_node(self, stats: np.ndarray, impurity: float, count: float) -> int:
        self.feature.append(-1)
        self.thr_bin.append(0)
        self.thr_raw.append(0.0)
        self.left.append(-1)
        self.right.append(-1)
        self.stats.append(stats)
        self.impurity.append(impurity)
        self.gain.append(0.0)
        self.count.append(count)
        return len(self.feature) - 1

    @property
    def n_nodes(self) -> int:
        return len(self.feature)

    def prediction(self, i: int) -> float:
        s = self.stats[i]
        if len(s) > 1:
            return float(np.argmax(s))
        return float(s[0])

    def depth(self) -> int:
        # compute via BFS
        if not self.feature:
            return 0
        depth = [0] * self.n_nodes
        best = 0
        for i in range(self.n_nodes):
            if self.feature[i] >= 0:
                for ch in (self.left[i], self.right[i]):
                    depth[ch] = depth[i] + 1
                    best = max(best, depth[ch])
        return best


def _feature_subset_size(strategy: str, F: int, is_classification: bool, num_trees: int) -> int:
    s = str(strategy)
    if s == "auto":
        if num_trees == 1:
            return F
        return max(1, int(math.sqrt(F))) if is_classification else max(1, F // 3)
    if s == "all":
        return F
    if s == "sqrt":
        return max(1, int(math.sqrt(F)))
    if s == "onethird":
        return max(1, F // 3)
    if s == "log2":
        return max(1, int(math.log2(F)))
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


def _counts_of(stats: torch.Tensor) -> torch.Tensor:
    if stats.shape[-1] == 3:
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
    nL = _counts_of(L)
    nR = _counts_of(R)
    n = _counts_of(parent).clamp_min(1e-12)  # G
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
    # threshold on usable gain
    ok = (best_gain > 0) & (best_gain >= min_info_gain) & torch.isfinite(best_gain)
    best_gain = torch.where(ok, best_gain, torch.zeros_like(best_gain))
    best_f = torch.where(ok, best_f, torch.full_like(best_f, -1))
    # left stats of the chosen split (G×S)
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


def _make_chunks(frontier, g_max: int, l_cap: int):
    """Split the (tree-major sorted) frontier into chunks obeying both the
    total-group budget (maxMemoryInMB) and the per-tree LDS cap."""
    chunks = []
    cur: List[Tuple[int, int, int]] = []
    per_tree = 0
    cur_tree = -1
    for item in frontier:
        t = item[0]
        if t != cur_tree:
            cur_tree = t
            per_tree = 0
        if cur and (len(cur) >= g_max or per_tree >= l_cap):
            chunks.append(cur)
            cur = []
            per_tree = 0
        cur.append(item)
        per_tree += 1
    if cur:
        chunks.append(cur)
    return chunks


def grow_trees(
    codes: torch.Tensor,  # N×F uint8 (local shard)
    labels_or_targets: torch.Tensor,  # N int (classification) / N float (regression)
    bag: Optional[torch.Tensor],  # T×N weights (Poisson) or None
    thresholds: torch.Tensor,  # F×(B-1) raw split values
    cfg: GrowthConfig,
    row_weights: Optional[torch.Tensor] = None,  # optional extra per-row weight
) -> List[FlatTree]:
    """Grow ``cfg.num_trees`` trees level-synchronously. Replicated across ranks:
    identical return on every rank (histograms are all-reduced)."""
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

    trees = [FlatTree() for _ in range(T)]
    nodeid = torch.zeros((T, N), dtype=torch.int32, device=dev)

    # root stats per tree: one K2 pass over feature 0 only (lut: node0 -> t)
    lut0 = torch.arange(T, dtype=torch.int32, device=dev).reshape(T, 1)
    tree_lo0 = torch.arange(T, dtype=torch.int32, device=dev)
    tree_cnt0 = torch.ones(T, dtype=torch.int32, device=dev)
    fs0 = torch.zeros((T, 1), dtype=torch.int32, device=dev)
    h0 = ops.tree_hist_nodes(
        codes, nodeid, lut0, tree_lo0, tree_cnt0, labels_u8, bag, T, 1, B,
        S if is_cls else 0, fs0, targets=targets,
    )
    dist.all_reduce_(h0.reshape(-1) if h0.is_contiguous() else h0)
    root_stats = h0.sum(dim=2)[:, 0, :]  # T×S
    rs_np = root_stats.detach().cpu().numpy().astype(np.float64)

    frontier: List[Tuple[int, int, int]] = []  # (tree, local node id, depth)
    for t in range(T):
        stats_t = rs_np[t]
        imp = float(_impurity_from_stats(torch.from_numpy(stats_t), cfg.impurity))
        cnt = float(stats_t.sum() if is_cls else stats_t[2])
        nid = trees[t].add_node(stats_t, imp, cnt)
        if cfg.max_depth > 0 and cnt >= 2 * cfg.min_instances_per_node and imp > 0:
            frontier.append((t, nid, 0))

    # memory budget → groups per histogram chunk (maxMemoryInMB parity) and
    # per-tree LDS cap (GPU kernel constraint)
    bytes_per_node = sub_F * B * S * 4
    g_max = max(1, int(cfg.max_memory_mb * 1024 * 1024 // max(bytes_per_node, 1)))
    l_cap = _hist_lds_cap(sub_F, B, S)

    thr_host = thresholds.cpu().numpy()
    level = 0
    while frontier and level < cfg.max_depth:
        next_frontier: List[Tuple[int, int, int]] = []
        for ci, chunk in enumerate(_make_chunks(frontier, g_max, l_cap)):
            t_chunk0 = _tick()
            G = len(chunk)
            chunk_t = np.fromiter((c[0] for c in chunk), dtype=np.int64, count=G)
            chunk_n = np.fromiter((c[1] for c in chunk), dtype=np.int64, count=G)
            chunk_d = np.fromiter((c[2] for c in chunk), dtype=np.int64, count=G)
            max_nodes = int(max(trees[t].n_nodes for t, _, _ in chunk))
            lut_np = np.full((T, max_nodes), -1, dtype=np.int32)
            lut_np[chunk_t, chunk_n] = np.arange(G, dtype=np.int32)
            per_tree_count = np.bincount(chunk_t, minlength=T).astype(np.int32)
            # first group index per tree (tree-major contiguity)
            tree_lo_np = np.zeros(T, dtype=np.int32)
            nz = per_tree_count > 0
            starts = np.cumsum(per_tree_count) - per_tree_count
            tree_lo_np[nz] = starts[nz]
            l_max = int(per_tree_count.max())
            lut_d = torch.from_numpy(lut_np).to(dev)
            tree_lo = torch.from_numpy(tree_lo_np).to(dev)
            tree_cnt = torch.from_numpy(per_tree_count).to(dev)
            # per-node feature subsets: one seeded RNG per (level, chunk) —
            # the frontier order is deterministic and identical on every rank,
            # so this is rank-invariant and reproducible (√F draw, K2 input)
            if use_subset:
                r = np.random.default_rng((cfg.seed * 1_000_003 + level * 8191 + ci) & 0x7FFFFFFF)
                keys = r.random((G, F))
                fs_np = np.sort(np.argpartition(keys, sub_F - 1, axis=1)[:, :sub_F], axis=1).astype(np.int32)
                fs = torch.from_numpy(fs_np).to(dev)
            else:
                fs_np = None
                fs = None
            _acc("chunk_setup", t_chunk0)
            t0 = _tick()
            hist = ops.tree_hist_nodes(
                codes, nodeid, lut_d, tree_lo, tree_cnt, labels_u8, bag, G, l_max, B,
                S if is_cls else 0, fs, targets=targets,
            )
            _acc("hist", t0)
            t0 = _tick()
            dist.all_reduce_(hist.reshape(-1))
            _acc("allreduce", t0)
            t0 = _tick()
            best_f, best_bin, best_gain, parent, left_stats = split_scan(
                hist, cfg.impurity, cfg.min_instances_per_node, cfg.min_info_gain
            )
            _acc("scan", t0)
            t0 = _tick()
            # vectorized bookkeeping: one host transfer per chunk
            right_stats = parent - left_stats
            imp_l_t = _impurity_from_stats(left_stats, cfg.impurity)
            imp_r_t = _impurity_from_stats(right_stats, cfg.impurity)
            cnt_l_t = _counts_of(left_stats)
            cnt_r_t = _counts_of(right_stats)
            packed = torch.cat(
                [
                    best_f.to(torch.float64).reshape(G, 1),
                    best_bin.to(torch.float64).reshape(G, 1),
                    best_gain.reshape(G, 1),
                    imp_l_t.reshape(G, 1),
                    imp_r_t.reshape(G, 1),
                    cnt_l_t.reshape(G, 1),
                    cnt_r_t.reshape(G, 1),
                    left_stats,
                    right_stats,
                ],
                dim=1,
            ).cpu().numpy()
            bf = packed[:, 0].astype(np.int64)
            bb = packed[:, 1].astype(np.int64)
            bg = packed[:, 2]
            il = packed[:, 3]
            ir = packed[:, 4]
            cl = packed[:, 5]
            cr = packed[:, 6]
            Sd = left_stats.shape[1]
            ls = packed[:, 7 : 7 + Sd]
            rs = packed[:, 7 + Sd : 7 + 2 * Sd]
            min_c = 2 * cfg.min_instances_per_node
            for g in range(G):
                if bf[g] < 0:
                    continue  # stays leaf
                t = int(chunk_t[g])
                nid = int(chunk_n[g])
                d = int(chunk_d[g])
                tr = trees[t]
                gf = int(fs_np[g, bf[g]]) if fs_np is not None else int(bf[g])
                lid = tr.add_node(ls[g], il[g], cl[g])
                rid = tr.add_node(rs[g], ir[g], cr[g])
                tr.feature[nid] = gf
                tr.thr_bin[nid] = int(bb[g])
                tr.thr_raw[nid] = float(thr_host[gf, bb[g]])
                tr.left[nid] = lid
                tr.right[nid] = rid
                tr.gain[nid] = float(bg[g])
                if d + 1 < cfg.max_depth:
                    if cl[g] >= min_c and il[g] > 0:
                        next_frontier.append((t, lid, d + 1))
                    if cr[g] >= min_c and ir[g] > 0:
                        next_frontier.append((t, rid, d + 1))
            _acc("bookkeep", t0)
        # advance node assignments one level, all trees in one launch (K4b);
        # skipped after the last split level (no further histogram pass reads it)
        if not next_frontier:
            break
        t0 = _tick()
        feat_cat = torch.tensor(
            [f for tr in trees for f in tr.feature], dtype=torch.int32, device=dev
        )
        thr_cat = torch.tensor(
            [x for tr in trees for x in tr.thr_bin], dtype=torch.int32, device=dev
        )
        left_cat = torch.tensor(
            [x for tr in trees for x in tr.left], dtype=torch.int32, device=dev
        )
        right_cat = torch.tensor(
            [x for tr in trees for x in tr.right], dtype=torch.int32, device=dev
        )
        ofs = [0]
        for tr in trees:
            ofs.append(ofs[-1] + tr.n_nodes)
        node_ofs = torch.tensor(ofs, dtype=torch.int32, device=dev)
        nodeid = ops.node_advance_multi(
            codes, nodeid, feat_cat, thr_cat, left_cat, right_cat, node_ofs
        )
        _acc("advance", t0)
        frontier = next_frontier
        level += 1

    if timing:
        print("grow_trees timing:", {k: round(v * 1000, 1) for k, v in tmr.items()})
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
            feature[ti, :n] = tr.feature
            threshold[ti, :n] = tr.thr_raw
            left[ti, :n] = tr.left
            right[ti, :n] = tr.right
            for i in range(n):
                s = tr.stats[i]
                if mode == "prob_sum":
                    tot = s.sum()
                    value[ti, i] = (s / tot) if tot > 0 else 0.0
                else:
                    # regression leaf: mean target
                    value[ti, i, 0] = s[0] / s[2] if s[2] > 0 else 0.0
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
            value = value * self._tree_weights.view(-1, 1, 1)
        return ops.tree_predict(
            X.to(torch.float32), self._feature, self._threshold, self._left, self._right, value
        )

    def feature_importances(self, num_features: int) -> np.ndarray:
        """Spark semantics: per tree, Σ_internal gain·count, normalized per
        tree, summed, normalized again (spark: TreeEnsembleModel.featureImportances)."""
        total = np.zeros(num_features, dtype=np.float64)
        for tr in self.trees:
            imp = np.zeros(num_features, dtype=np.float64)
            for i in range(tr.n_nodes):
                f = tr.feature[i]
                if f >= 0:
                    imp[f] += tr.gain[i] * tr.count[i]
            s = imp.sum()
            if s > 0:
                total += imp / s
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
