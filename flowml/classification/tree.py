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

Categorical features (declared via the features column's metadata — set by
VectorIndexerModel, or by VectorAssembler from nominal input columns) use
Spark's categorical split-sets: identity binning (bin = category), ordered
centroid-sort scan or unordered subset enumeration in the split search
(``binsToBestSplit`` categorical branches), and set-membership routing in
advance/inference. Arity is capped at min(maxBins, 63) (int64 masks).
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
    # per-feature categorical arity (len-F int array; 0 = continuous).
    # Categorical features use identity binning (bin = category) and Spark's
    # categorical split-set semantics in the scan.
    cat_arity: Optional[np.ndarray] = None


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
        # categorical split-set per node: bit c of cat_mask set = category c
        # goes LEFT; 0 = continuous split (or leaf). num_cats = feature arity
        # at a categorical split (0 otherwise) — persisted as numCategories.
        self._cat_mask = np.zeros(cap, np.int64)
        self._num_cats = np.zeros(cap, np.int32)

    def _ensure(self, extra: int) -> None:
        need = self._n + extra
        if need <= self._cap:
            return
        newcap = max(need, self._cap * 2)
        old = (self._feature, self._thr_bin, self._thr_raw, self._left,
               self._right, self._impurity, self._gain, self._count, self._stats,
               self._cat_mask, self._num_cats)
        n = self._n
        self._cap = newcap
        self._alloc(newcap)
        for dst, src in zip(
            (self._feature, self._thr_bin, self._thr_raw, self._left,
             self._right, self._impurity, self._gain, self._count, self._stats,
             self._cat_mask, self._num_cats),
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

    @property
    def cat_mask(self) -> np.ndarray:
        return self._cat_mask[: self._n]

    @property
    def num_cats(self) -> np.ndarray:
        return self._num_cats[: self._n]

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

    @classmethod
    def from_arrays(cls, feature, thr_bin, thr_raw, left, right, impurity,
                    gain, count, stats, cat_mask, num_cats,
                    regression: bool = False) -> "FlatTree":
        """Adopt pre-built per-node arrays WITHOUT copying (they may be
        views into a shared arena gather). Trees built this way are
        read-only by convention — the grower never appends post-extract."""
        t = cls.__new__(cls)
        n = len(feature)
        t._n = n
        t._cap = n
        t._sw = stats.shape[1] if stats.ndim == 2 else 0
        t.regression = bool(regression)
        t._feature = feature
        t._thr_bin = thr_bin
        t._thr_raw = thr_raw
        t._left = left
        t._right = right
        t._impurity = impurity
        t._gain = gain
        t._count = count
        t._stats = stats
        t._cat_mask = cat_mask
        t._num_cats = num_cats
        return t

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

    def set_splits(self, nids, feats, bins, raws, lefts, rights, gains,
                   cat_masks=None, num_cats=None) -> None:
        self._feature[nids] = feats
        self._thr_bin[nids] = bins
        self._thr_raw[nids] = raws
        self._left[nids] = lefts
        self._right[nids] = rights
        self._gain[nids] = gains
        if cat_masks is not None:
            self._cat_mask[nids] = cat_masks
            self._num_cats[nids] = num_cats

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


def tree_debug_string(tree: "FlatTree", indentation: str = "  ") -> str:
    """Spark-style toDebugString body: If/Else lines per internal node,
    Predict lines per leaf (spark Node.subtreeToString)."""
    lines: List[str] = []

    def visit(i: int, depth: int):
        pad = indentation * depth
        f = int(tree.feature[i])
        if f < 0:
            lines.append(f"{pad}Predict: {tree.prediction(i)}")
            return
        if tree.cat_mask[i] != 0:
            cats = [c for c in range(64) if (int(tree.cat_mask[i]) >> c) & 1]
            cond = f"feature {f} in {{{','.join(str(float(c)) for c in cats)}}}"
            ncond = f"feature {f} not in {{{','.join(str(float(c)) for c in cats)}}}"
        else:
            thr = float(tree.thr_raw[i])
            cond = f"feature {f} <= {thr}"
            ncond = f"feature {f} > {thr}"
        lines.append(f"{pad}If ({cond})")
        visit(int(tree.left[i]), depth + 1)
        lines.append(f"{pad}Else ({ncond})")
        visit(int(tree.right[i]), depth + 1)

    if tree.n_nodes:
        visit(0, 1)
    return "\n".join(lines)


def ensemble_debug_string(header: str, trees, weights=None) -> str:
    parts = [header]
    for i, t in enumerate(trees):
        w = f" (weight {weights[i]})" if weights is not None else ""
        parts.append(f"  Tree {i}{w}:")
        body = tree_debug_string(t, indentation="   ")
        parts.append(body)
    return "\n".join(parts)


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
# Categorical split-sets (spark:.../tree/impl/RandomForest.binsToBestSplit,
# categorical branches; SURVEY.md §2.2 RF row / VERDICT r1 next-round #4)
# ---------------------------------------------------------------------------


def cat_arity_from_meta(dataset, features_col: str, F: int) -> Optional[np.ndarray]:
    """Per-feature arity array from the features column's metadata
    ({"categorical": {featIdx: arity}}, set by VectorIndexerModel or by
    VectorAssembler from nominal input columns — the stand-in for Spark's
    ML attribute groups). None when no feature is categorical."""
    m = dataset.meta(features_col) if hasattr(dataset, "meta") else {}
    cats = m.get("categorical")
    if not cats:
        return None
    arr = np.zeros(F, np.int64)
    for k, v in cats.items():
        k, v = int(k), int(v)
        if not (0 <= k < F):
            raise ValueError(f"categorical feature index {k} out of range [0, {F})")
        if v < 2:
            raise ValueError(f"categorical feature {k} has arity {v} (< 2)")
        arr[k] = v
    return arr


def max_unordered_arity(max_bins: int) -> int:
    """Spark DecisionTreeMetadata.maxCategoriesForUnorderedFeature:
    arity k is 'unordered' (full subset enumeration) iff
    2·(2^(k−1) − 1) ≤ maxBins, i.e. k ≤ floor(log2(maxBins/2 + 1)) + 1."""
    return int(math.floor(math.log2(max_bins / 2 + 1))) + 1 if max_bins >= 2 else 1


def apply_categorical_codes(
    codes: torch.Tensor, X: torch.Tensor, cat_arity: np.ndarray
) -> torch.Tensor:
    """Overwrite categorical feature columns of the binned code matrix with
    the raw category index (identity binning — TreePoint semantics for
    categorical features: bin = category). Values are validated to be
    integral in [0, arity)."""
    idx = np.nonzero(cat_arity > 0)[0]
    if not len(idx):
        return codes
    dev = X.device
    cols = torch.from_numpy(idx).to(dev)
    vals = X.index_select(1, cols)
    arity_t = torch.from_numpy(cat_arity[idx].astype(np.float32)).to(dev)
    bad = (vals < 0) | (vals >= arity_t) | (vals != vals.floor())
    if bool(bad.any()):
        j = int(torch.nonzero(bad)[0, 1])
        raise ValueError(
            f"categorical feature {int(idx[j])} has values outside "
            f"[0, {int(cat_arity[idx[j]])}) or non-integral values"
        )
    codes[:, cols] = vals.to(codes.dtype)
    return codes


def _categorical_best_splits(
    hist: torch.Tensor,      # G×Fp×B×S float64-able (cls) or G×Fp×B×2 (reg cells)
    slot_feat: np.ndarray,   # G×Fp global feature id per slot
    cat_arity: np.ndarray,   # F arity per feature (0 = continuous)
    cfg: "GrowthConfig",
):
    """Best categorical split per node over its categorical feature slots.

    Spark semantics: ordered-categorical features (binary classification,
    regression, or high-arity multiclass) sort categories by centroid —
    P(class 1), target mean, or label-distribution impurity respectively,
    empty categories last — and scan the arity−1 prefix splits (Breiman's
    trick: exact for binary/regression, heuristic for multiclass);
    low-arity multiclass features are 'unordered': all 2^(k−1)−1 subset
    splits are enumerated. Returns per-node numpy arrays
    (slot, gain, mask, arity, lstats, rstats, impL, impR, cntL, cntR);
    slot = −1 / gain = −inf where no categorical candidate exists.
    """
    dev = hist.device
    G, Fp = hist.shape[0], hist.shape[1]
    Sw = hist.shape[3]
    is_cls = cfg.is_classification
    imp_kind = cfg.impurity
    min_inst = cfg.min_instances_per_node
    arity_slot = cat_arity[slot_feat]  # G×Fp
    # per-position candidate buffers (first-max-wins merge per node at the end)
    gain_pos = np.full((G, Fp), -np.inf)
    mask_pos = np.zeros((G, Fp), np.int64)
    ls_pos = np.zeros((G, Fp, Sw))
    rs_pos = np.zeros((G, Fp, Sw))
    il_pos = np.zeros((G, Fp))
    ir_pos = np.zeros((G, Fp))
    cl_pos = np.zeros((G, Fp))
    cr_pos = np.zeros((G, Fp))
    unord_max = max_unordered_arity(cfg.max_bins)

    def _gain_cls(Lst, parent):
        # Lst: M×C×S prefix/subset left stats; parent: M×S
        Rst = parent.unsqueeze(1) - Lst
        nL = Lst.sum(-1)
        nR = Rst.sum(-1)
        n = parent.sum(-1).clamp_min(1e-12)
        iL = _impurity_from_stats(Lst, imp_kind)
        iR = _impurity_from_stats(Rst, imp_kind)
        iP = _impurity_from_stats(parent, imp_kind)
        g = iP.unsqueeze(1) - (nL * iL + nR * iR) / n.unsqueeze(1)
        valid = (nL >= min_inst) & (nR >= min_inst)
        return torch.where(valid, g, torch.full_like(g, -float("inf"))), Rst, iL, iR, nL, nR

    def _gain_reg(Lst, parent):
        # stats are (Σwt, Σw); canonical variance-gain formula — identical
        # op order to split_scan_packed_reg so merged gains are comparable
        Rst = parent.unsqueeze(1) - Lst
        sL, nL = Lst[..., 0], Lst[..., 1]
        sR, nR = Rst[..., 0], Rst[..., 1]
        sP, nP = parent[..., 0], parent[..., 1].clamp_min(1e-12)
        g = (sL * sL / nL.clamp_min(1e-300) + sR * sR / nR.clamp_min(1e-300)
             - (sP * sP / nP).unsqueeze(1)) / nP.unsqueeze(1)
        valid = (nL >= min_inst) & (nR >= min_inst)
        return torch.where(valid, g, torch.full_like(g, -float("inf"))), Rst, None, None, nL, nR

    H = hist.to(torch.float64)
    for k in np.unique(arity_slot[arity_slot > 0]):
        k = int(k)
        gi_np, si_np = np.nonzero(arity_slot == k)
        if not len(gi_np):
            continue
        gi = torch.from_numpy(gi_np).to(dev)
        si = torch.from_numpy(si_np).to(dev)
        Hs = H[gi, si][:, :k, :]  # M×k×S
        parent = Hs.sum(1)        # M×S
        M = Hs.shape[0]
        unordered = is_cls and cfg.n_classes > 2 and k <= unord_max
        if unordered and k >= 2:
            n_masks = (1 << (k - 1)) - 1
            m_t = torch.arange(1, n_masks + 1, dtype=torch.int64, device=dev)
            bits = ((m_t.unsqueeze(1) >> torch.arange(k, device=dev)) & 1).to(
                torch.float64
            )  # n_masks×k (bit k−1 always 0: complement symmetry)
            Lst = torch.einsum("mk,nks->nms", bits, Hs)  # M×n_masks×S
            g, Rst, iL, iR, nL, nR = _gain_cls(Lst, parent)
            bi = g.argmax(dim=1)  # first max wins
            ar = torch.arange(M, device=dev)
            best_g = g[ar, bi]
            best_mask = m_t[bi]
        else:
            cnt = Hs.sum(-1) if is_cls else Hs[..., 1]
            if not is_cls:
                centroid = Hs[..., 0] / cnt.clamp_min(1e-300)
            elif cfg.n_classes == 2:
                centroid = Hs[..., 1] / cnt.clamp_min(1e-300)
            else:
                centroid = _impurity_from_stats(Hs, imp_kind)
            centroid = torch.where(
                cnt > 0, centroid, torch.full_like(centroid, float("inf"))
            )
            order = torch.argsort(centroid, dim=1, stable=True)  # M×k
            Hsort = Hs.gather(1, order.unsqueeze(-1).expand(-1, -1, Sw))
            pref = Hsort.cumsum(dim=1)[:, : k - 1, :]  # M×(k−1)×S prefixes
            gain_fn = _gain_cls if is_cls else _gain_reg
            g, Rst, iL, iR, nL, nR = gain_fn(pref, parent)
            bi = g.argmax(dim=1)
            ar = torch.arange(M, device=dev)
            best_g = g[ar, bi]
            # mask of the winning prefix: OR of bits of order[:, :bi+1]
            bitvals = (torch.ones(1, dtype=torch.int64, device=dev) << order)
            maskpref = bitvals.cumsum(dim=1)  # exact: bits are disjoint
            best_mask = maskpref[ar, bi]
            Lst = pref
        best_ls = Lst[ar, bi]
        best_rs = Rst[ar, bi]
        gain_pos[gi_np, si_np] = best_g.cpu().numpy()
        mask_pos[gi_np, si_np] = best_mask.cpu().numpy()
        ls_pos[gi_np, si_np] = best_ls.cpu().numpy()
        rs_pos[gi_np, si_np] = best_rs.cpu().numpy()
        if is_cls:
            il_pos[gi_np, si_np] = iL[ar, bi].cpu().numpy()
            ir_pos[gi_np, si_np] = iR[ar, bi].cpu().numpy()
        cl_pos[gi_np, si_np] = nL[ar, bi].cpu().numpy()
        cr_pos[gi_np, si_np] = nR[ar, bi].cpu().numpy()

    best_slot = gain_pos.argmax(axis=1)  # first max wins (numpy argmax)
    ar_np = np.arange(G)
    best_gain = gain_pos[ar_np, best_slot]
    none = ~np.isfinite(best_gain)
    best_slot = np.where(none, -1, best_slot)
    return {
        "slot": best_slot.astype(np.int64),
        "gain": best_gain,
        "mask": mask_pos[ar_np, np.maximum(best_slot, 0)],
        "arity": arity_slot[ar_np, np.maximum(best_slot, 0)].astype(np.int32),
        "ls": ls_pos[ar_np, np.maximum(best_slot, 0)],
        "rs": rs_pos[ar_np, np.maximum(best_slot, 0)],
        "il": il_pos[ar_np, np.maximum(best_slot, 0)],
        "ir": ir_pos[ar_np, np.maximum(best_slot, 0)],
        "cl": cl_pos[ar_np, np.maximum(best_slot, 0)],
        "cr": cr_pos[ar_np, np.maximum(best_slot, 0)],
    }


def _neutralize_cat_slots(hist: torch.Tensor, slot_feat: np.ndarray,
                          cat_arity: np.ndarray) -> None:
    """In place: collapse each categorical slot's histogram to bin 0 so the
    continuous scan (HIP K3 or torch) sees correct parent/self stats but
    never emits a split there (every prefix has an empty right side)."""
    arity_slot = cat_arity[slot_feat]
    gi_np, si_np = np.nonzero(arity_slot > 0)
    if not len(gi_np):
        return
    dev = hist.device
    gi = torch.from_numpy(gi_np).to(dev)
    si = torch.from_numpy(si_np).to(dev)
    tot = hist[gi, si].sum(dim=1)  # M×S
    hist[gi, si] = 0.0
    hist[gi, si, 0] = tot


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
    cat_arity_np = None
    if cfg.cat_arity is not None and (np.asarray(cfg.cat_arity) > 0).any():
        cat_arity_np = np.asarray(cfg.cat_arity, dtype=np.int64)
        amax = int(cat_arity_np.max())
        if amax > min(cfg.max_bins, 63):
            raise ValueError(
                f"categorical feature arity {amax} exceeds "
                f"min(maxBins={cfg.max_bins}, 63)"
            )
    all_feats_np = np.broadcast_to(
        np.arange(F, dtype=np.int64), (1, F)
    )  # slot map when no feature subsetting

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
            cat_best = None
            if cat_arity_np is not None:
                sf_np = fs_np if fs_np is not None else np.broadcast_to(
                    all_feats_np, (G, F))
                if is_cls:
                    cat_best = _categorical_best_splits(
                        hist, sf_np, cat_arity_np, cfg)
                    _neutralize_cat_slots(hist, sf_np, cat_arity_np)
                else:
                    cells_v = ops.reg_hist_views(hist, G, sub_F, B)[0]
                    cat_best = _categorical_best_splits(
                        cells_v, sf_np, cat_arity_np, cfg)
                    _neutralize_cat_slots(cells_v, sf_np, cat_arity_np)
            _acc("cat_scan", t0)
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
            cat_mask_arr = np.zeros(G, np.int64)
            cat_ncat_arr = np.zeros(G, np.int32)
            if cat_best is not None:
                # merge: first-max-wins across ALL slots — categorical wins on
                # strictly greater gain, or equal gain at a smaller slot index
                cg = cat_best["gain"]
                c_ok = np.isfinite(cg) & (cg > 0) & (cg >= cfg.min_info_gain)
                cont_gain = np.where(bf >= 0, bg, -np.inf)
                take = c_ok & (
                    (cg > cont_gain)
                    | ((cg == cont_gain) & (cat_best["slot"] < bf))
                )
                if take.any():
                    tk = np.nonzero(take)[0]
                    bf[tk] = cat_best["slot"][tk]
                    bb[tk] = 0
                    bg[tk] = cg[tk]
                    cat_mask_arr[tk] = cat_best["mask"][tk]
                    cat_ncat_arr[tk] = cat_best["arity"][tk]
                    cl[tk] = cat_best["cl"][tk]
                    cr[tk] = cat_best["cr"][tk]
                    if is_cls:
                        il[tk] = cat_best["il"][tk]
                        ir[tk] = cat_best["ir"][tk]
                        ls[tk] = cat_best["ls"][tk]
                        rs[tk] = cat_best["rs"][tk]
                    else:
                        # cat reg stats are (Σwt, Σw) → map into (s, s², n)
                        ls[tk] = 0.0
                        rs[tk] = 0.0
                        ls[tk, 0] = cat_best["ls"][tk, 0]
                        ls[tk, 2] = cat_best["ls"][tk, 1]
                        rs[tk, 0] = cat_best["rs"][tk, 0]
                        rs[tk, 2] = cat_best["rs"][tk, 1]
            sel = np.nonzero(bf >= 0)[0]
            if len(sel):
                k = len(sel)
                gf = fs_np[sel, bf[sel]] if fs_np is not None else bf[sel]
                raws = thr_host[gf, np.maximum(bb[sel], 0)]
                sel_mask = cat_mask_arr[sel]
                sel_ncat = cat_ncat_arr[sel]
                is_cat_sel = sel_mask != 0
                if is_cat_sel.any():
                    raws = raws.copy()
                    raws[is_cat_sel] = np.nan  # categorical: mask routes, not thr
                    bb_sel = bb[sel].copy()
                    bb_sel[is_cat_sel] = -1
                else:
                    bb_sel = bb[sel]
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
                arena.set_splits(chunk_n[sel], gf, bb_sel, raws, lid, rid, bg[sel],
                                 cat_masks=sel_mask, num_cats=sel_ncat)
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
            torch.from_numpy(arena.cat_mask.copy()).to(dev),
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
            torch.from_numpy(arena.cat_mask.copy()).to(dev),
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
    catm_o = arena.cat_mask[order]
    ncat_o = arena.num_cats[order]
    internal_o = feat_o >= 0
    left_o = np.where(internal_o, g2l[np.where(internal_o, arena.left[order], 0)], -1)
    right_o = np.where(internal_o, g2l[np.where(internal_o, arena.right[order], 0)], -1)
    trees: List[FlatTree] = []
    for t in range(T):
        sl = slice(bounds[t], bounds[t + 1])
        # adopt contiguous slices of the one arena gather as views — no
        # per-tree copies (was ~0.5 ms of the flagship fit across 100 trees)
        trees.append(FlatTree.from_arrays(
            feat_o[sl], thb_o[sl], thr_o[sl], left_o[sl], right_o[sl],
            imp_o[sl], gain_o[sl], cnt_o[sl], st_o[sl], catm_o[sl],
            ncat_o[sl], regression=not is_cls,
        ))
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
        catmask = np.zeros((T, M), dtype=np.int64)
        value = np.zeros((T, M, K), dtype=np.float32)
        for ti, tr in enumerate(trees):
            n = tr.n_nodes
            if n == 0:
                continue
            feature[ti, :n] = tr.feature
            threshold[ti, :n] = tr.thr_raw
            left[ti, :n] = tr.left
            right[ti, :n] = tr.right
            catmask[ti, :n] = tr.cat_mask
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
        self._catmask = torch.from_numpy(catmask)
        self._value = torch.from_numpy(value)
        self._packed = ops.pack_tree_nodes(
            self._feature, self._threshold, self._left, self._right,
            self._catmask)
        self._tree_weights: Optional[torch.Tensor] = None  # GBT stepSize weights

    def set_tree_weights(self, w: Sequence[float]) -> None:
        self._tree_weights = torch.tensor(list(w), dtype=torch.float32)

    def to(self, device) -> "TreeEnsemble":
        self._feature = self._feature.to(device)
        self._threshold = self._threshold.to(device)
        self._left = self._left.to(device)
        self._right = self._right.to(device)
        self._catmask = self._catmask.to(device)
        self._value = self._value.to(device)
        self._packed = self._packed.to(device)
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
            X.to(torch.float32), self._feature, self._threshold, self._left,
            self._right, value, self._catmask, packed=self._packed
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
