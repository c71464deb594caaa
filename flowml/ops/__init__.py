"""Compute ops: HIP/CDNA4 kernels with torch reference fallbacks.

Dispatch policy (SURVEY.md §2.4, §7 "No-GPU dev loop"):

* On a GPU (ROCm) box the hand-written gfx950 HIP kernels in
  ``flowml/kernels/`` are REQUIRED for the hot ops — if the compiled
  extension is missing, ops raise rather than silently falling back to eager
  torch (the silent-fallback failure mode called out in the task brief).
  ``FLOWML_FORCE_TORCH_OPS=1`` exists for A/B benchmarking only.
* On CPU (this dev sandbox, unit tests) the torch/numpy reference
  implementations run; every kernel has a bitwise (integer histograms) or
  tolerance (GEMM) parity test against them (SURVEY.md §4 "Kernel parity").

Data layouts are chosen for the GPU kernels and shared by the CPU reference:
per-(row,tree) state (node ids, bag weights) is **T-major** ``(T, N)`` so a
block working tree ``t`` reads coalesced rows; binned codes are row-major
``(N, F)`` uint8 (41 B/row at NSL-KDD shape — the only per-row bytes the
flagship hot loop touches).
"""

from __future__ import annotations

import os
from typing import List, Optional, Tuple

import torch

_ext = None
_ext_err: Optional[str] = None


def _load_ext():
    """Load the compiled HIP extension (in-tree .so). Cached."""
    global _ext, _ext_err
    if _ext is not None or _ext_err is not None:
        return _ext
    try:
        from flowml.kernels import load_extension

        _ext = load_extension()
    except Exception as e:  # noqa: BLE001
        _ext_err = f"{type(e).__name__}: {e}"
        _ext = None
    return _ext


def force_torch() -> bool:
    return os.environ.get("FLOWML_FORCE_TORCH_OPS", "0") == "1"


def hip_available(t: torch.Tensor) -> bool:
    """True when the HIP kernel path must be used for this tensor."""
    if not t.is_cuda or force_torch():
        return False
    ext = _load_ext()
    if ext is None:
        raise RuntimeError(
            "flowml HIP kernel extension is not built but a GPU tensor reached "
            f"the ops layer (load error: {_ext_err}). Build it with "
            "`python -m flowml.kernels.build` — silent eager fallback on GPU "
            "is disabled by design."
        )
    return True


def get_ext():
    return _load_ext()


def _empty(device, dtype=torch.float32):
    return torch.empty(0, dtype=dtype, device=device)


# ---------------------------------------------------------------------------
# K4a — quantile binning: codes = searchsorted(thresholds, x)
# ---------------------------------------------------------------------------


def bin_codes(X: torch.Tensor, thresholds: torch.Tensor) -> torch.Tensor:
    """Quantize N×F float features to uint8 bin codes given per-feature
    threshold rows (F×(B-1), +inf padded). code = #(thresholds < x), strict —
    so the binned split "code <= b" equals the raw rule "x <= thr[b]" used at
    inference (K8), exactly, even for integer-coded features.

    Reference site: TreePoint.convertToTreeRDD.
    """
    if hip_available(X):
        return get_ext().bin_codes(X.contiguous(), thresholds.contiguous())
    return _bin_codes_torch(X, thresholds)


def _bin_codes_torch(X, thresholds):
    codes = torch.searchsorted(
        thresholds.to(X.dtype), X.T.contiguous(), right=False
    ).T
    return codes.to(torch.uint8).contiguous()


# ---------------------------------------------------------------------------
# K2 — per-node histograms, all trees in one pass (the flagship hot loop)
# ---------------------------------------------------------------------------


def poisson_cdf_table(rate: float, mode: str) -> torch.Tensor:
    """8-entry float32 inverse-CDF table for the in-kernel hash bagging.
    Poisson: cdf[k] = P(X<=k); Bernoulli: cdf[0] = rate, rest 2.0 (never hit)."""
    import math

    if mode == "bernoulli":
        t = [float(rate)] + [2.0] * 7
    else:
        acc = 0.0
        t = []
        term = math.exp(-rate)
        for k in range(8):
            acc += term
            t.append(min(acc, 1.0))
            term *= rate / (k + 1)
    return torch.tensor(t, dtype=torch.float32)


def _hash_bag_torch(mode: int, cdf: torch.Tensor, seed: int, grow_ids, t: int):
    """Bitwise-identical CPU reference of the kernel's splitmix64 bag weight."""
    import numpy as np

    g = (np.asarray(grow_ids, dtype=np.uint64) + np.uint64(1))
    with np.errstate(over="ignore"):
        x = (np.uint64(seed)
             ^ (np.uint64(0x9E3779B97F4A7C15) * g)
             ^ (np.uint64(0xBF58476D1CE4E5B9) * np.uint64(t + 1)))
        x = x + np.uint64(0x9E3779B97F4A7C15)
        x = (x ^ (x >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        x = (x ^ (x >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        x = x ^ (x >> np.uint64(31))
    u = (x >> np.uint64(40)).astype(np.float32) * np.float32(1.0 / 16777216.0)
    c = cdf.cpu().numpy()
    if mode == 3:
        return (u < c[0]).astype(np.float32)
    k = np.zeros(u.shape, dtype=np.int32)
    for i in range(8):
        k += (u >= c[i]).astype(np.int32)
    return k.astype(np.float32)


def tree_hist_nodes(
    codes: torch.Tensor,   # N×F uint8
    nodeid: torch.Tensor,  # T×N int32 GLOBAL arena node id per (tree,row), -1 done
    lut: torch.Tensor,     # n_nodes int32: arena id -> chunk group index, -1
    tree_lo: torch.Tensor, # T int32: first group index of tree t in the chunk
    tree_cnt: torch.Tensor,  # T int32: number of groups of tree t in the chunk
    labels: torch.Tensor,  # N uint8 class labels (classification)
    bag: Optional[torch.Tensor],      # T×N float32 bag weights (None = 1)
    n_groups: int,
    l_max: int,            # max groups of any single tree in this chunk
    n_bins: int,
    n_classes: int,
    feat_subset: Optional[torch.Tensor] = None,  # G×F' int32
    targets: Optional[torch.Tensor] = None,      # N float32 (regression)
    advance=None,          # (feat, thr_bin, left, right) — global arena arrays
    hash_bag=None,         # (mode: 2=poisson|3=bernoulli, cdf 8×f32, seed, row_offset)
) -> torch.Tensor:
    """Classification: out[g,f',b,c] += w  (G×F'×B×C float32).
    Regression (targets given): FLAT float32 of length G·F'·B·2 + G —
    cells out[g,f',b,(Σwt, Σw)] followed by per-node Σwt² slots (variance
    split gain needs no per-bin Σt²; see tree_kernels.hip K2b).
    Use ``reg_hist_views`` to split the flat tensor.

    ``advance``: fused K4b — each (tree,row) is advanced one level down its
    tree (``nodeid`` is updated IN PLACE) before accumulation; pass it on the
    FIRST chunk of each level. ``hash_bag``: in-kernel Poisson/Bernoulli
    BaggedPoint weights keyed by (seed, global row, tree) — no T×N bag tensor.

    Reference site: DTStatsAggregator bin update inside findBestSplits
    (SURVEY.md §2.4 K2). The GPU kernel privatizes one tree's frontier
    histograms in LDS per workgroup and flushes with global atomics.
    """
    dev = codes.device
    if hip_available(codes):
        ext = get_ext()
        fs = feat_subset if feat_subset is not None else _empty(dev, torch.int32)
        bg = bag if bag is not None else _empty(dev)
        if advance is not None:
            af, at, al, ar, am = advance
        else:
            e = _empty(dev, torch.int32)
            af = at = al = ar = e
            am = _empty(dev, torch.int64)
        if hash_bag is not None:
            bm, cdf, bseed, rofs = hash_bag
            cdf = cdf.to(dev)
        else:
            bm, cdf, bseed, rofs = 0, _empty(dev), 0, 0
        if targets is None:
            return ext.tree_hist_cls(
                codes, nodeid, lut, tree_lo, tree_cnt, labels, bg,
                int(n_groups), int(l_max), int(n_bins), int(n_classes), fs,
                af, at, al, ar, am, int(bm), cdf, int(bseed), int(rofs),
            )
        return ext.tree_hist_reg(
            codes, nodeid, lut, tree_lo, tree_cnt, targets, bg,
            int(n_groups), int(l_max), int(n_bins), fs,
            af, at, al, ar, am, int(bm), cdf, int(bseed), int(rofs),
        )
    if advance is not None:
        af, at, al, ar, am = advance
        nodeid.copy_(node_advance(codes, nodeid, af, at, al, ar, am))
    return _tree_hist_nodes_torch(
        codes, nodeid, lut, labels, bag, n_groups, n_bins, n_classes,
        feat_subset, targets, hash_bag,
    )


def _tree_hist_nodes_torch(codes, nodeid, lut, labels, bag, n_groups, n_bins,
                           n_classes, feat_subset, targets, hash_bag=None):
    N, F = codes.shape
    T = nodeid.shape[0]
    dev = codes.device
    stat_w = 2 if targets is not None else n_classes
    Fp = feat_subset.shape[1] if (feat_subset is not None and feat_subset.numel()) else F
    out = torch.zeros(n_groups * Fp * n_bins * stat_w, dtype=torch.float64, device=dev)
    ns2 = torch.zeros(n_groups, dtype=torch.float64, device=dev)
    lut64 = lut.to(torch.int64)
    for t in range(T):
        nid = nodeid[t].to(torch.int64)
        live = nid >= 0
        g_t = torch.where(live, lut64[nid.clamp_min(0)], nid.new_full((N,), -1))
        active = g_t >= 0
        hw_t = None
        if hash_bag is not None:
            bm, cdf, bseed, rofs = hash_bag
            import numpy as _np

            hw_t = torch.from_numpy(
                _hash_bag_torch(bm, cdf, bseed, _np.arange(rofs, rofs + N), t)
            ).to(dev)
            active = active & (hw_t > 0)
        if bag is not None:
            w_t = bag[t].to(torch.float64)
            active = active & (w_t > 0)
        if not bool(active.any()):
            continue
        rows = torch.nonzero(active, as_tuple=False).flatten()
        g_a = g_t[rows]
        c_a = codes[rows].to(torch.int64)
        w_a = (bag[t][rows].to(torch.float64) if bag is not None
               else torch.ones(len(rows), dtype=torch.float64, device=dev))
        if hw_t is not None:
            w_a = w_a * hw_t[rows].to(torch.float64)
        if feat_subset is not None and feat_subset.numel():
            feats = feat_subset.to(torch.int64)[g_a]
            c_a = c_a.gather(1, feats)
        f = torch.arange(Fp, dtype=torch.int64, device=dev).unsqueeze(0)
        base = ((g_a.unsqueeze(1) * Fp + f) * n_bins + c_a) * stat_w
        if targets is None:
            lab = labels[rows].to(torch.int64).unsqueeze(1)
            out.scatter_add_(0, (base + lab).reshape(-1),
                             w_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
        else:
            tv = targets[rows].to(torch.float64)
            out.scatter_add_(0, base.reshape(-1),
                             (w_a * tv).unsqueeze(1).expand(-1, Fp).reshape(-1))
            out.scatter_add_(0, (base + 1).reshape(-1),
                             w_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
            ns2.scatter_add_(0, g_a, w_a * tv * tv)
    if targets is not None:
        # flat [cells | node_s2] — the layout the HIP kernel writes
        return torch.cat([out, ns2]).to(torch.float32)
    return out.reshape(n_groups, Fp, n_bins, stat_w).to(torch.float32)


def split_scan_packed_reg(
    flat_hist: torch.Tensor,  # G·Fp·B·2 + G flat (already all-reduced)
    G: int, Fp: int, B: int,
    min_instances: int,
    min_info_gain: float,
) -> torch.Tensor:
    """K3 variance scan on the 2-wide regression layout. Per-node packed row
    (G×10 float32): [best_f, best_bin, gain, sL, nL, sR, nR, imp_self,
    s_self, n_self]. Children impurities are NOT computable here (no per-bin
    Σt²) — the grower backfills leaf impurities from the per-leaf stats pass.
    Canonical gain (identical op order on both paths, all f64):
        gain = (sL·sL/nL + sR·sR/nR − sP·sP/nP) / nP
    which equals Spark's imp_p − wL·imp_L − wR·imp_R algebraically.
    """
    if hip_available(flat_hist):
        return get_ext().split_scan_reg(
            flat_hist.contiguous(), int(G), int(Fp), int(B),
            int(min_instances), float(min_info_gain)
        )
    cells, ns2 = reg_hist_views(flat_hist, G, Fp, B)
    sb = cells[..., 0].to(torch.float64)
    nb = cells[..., 1].to(torch.float64)
    sP = sb[:, 0, :].sum(-1)                       # G
    nP_raw = nb[:, 0, :].sum(-1)
    nP = nP_raw.clamp_min(1e-12)
    s2P = ns2.to(torch.float64)
    m = sP / nP
    imp_self = (s2P / nP - m * m).clamp_min(0.0)
    sl = sb.cumsum(-1)[..., : B - 1]               # G,Fp,B-1
    cl = nb.cumsum(-1)[..., : B - 1]
    sr = sb.sum(-1, keepdim=True) - sl
    cr = nb.sum(-1, keepdim=True) - cl
    valid = (cl >= min_instances) & (cr >= min_instances)
    clc = cl.clamp_min(1e-300)
    crc = cr.clamp_min(1e-300)
    gP = sP * sP / nP
    gain = (sl * sl / clc + sr * sr / crc - gP.reshape(G, 1, 1)) / nP.reshape(G, 1, 1)
    gain = torch.where(valid, gain, torch.full_like(gain, -float("inf")))
    flat = gain.reshape(G, -1)
    idx = flat.argmax(dim=1)                       # first max wins ties
    bg = flat.gather(1, idx.unsqueeze(1)).squeeze(1)
    bf = idx // (B - 1)
    bb = idx % (B - 1)
    ok = torch.isfinite(bg) & (bg > 0) & (bg >= min_info_gain)
    ar = torch.arange(G, device=flat_hist.device)
    slb = sl[ar, bf, bb]
    clb = cl[ar, bf, bb]
    srb = sr[ar, bf, bb]
    crb = cr[ar, bf, bb]
    z = torch.zeros_like(bg)
    out = torch.stack([
        torch.where(ok, bf.to(torch.float64), torch.full_like(bg, -1.0)),
        torch.where(ok, bb.to(torch.float64), z),
        torch.where(ok, bg, z),
        torch.where(ok, slb, z),
        torch.where(ok, clb, z),
        torch.where(ok, srb, z),
        torch.where(ok, crb, z),
        imp_self,
        sP,
        nP_raw,
    ], dim=1)
    return out.to(torch.float32)


def reg_hist_views(flat: torch.Tensor, G: int, Fp: int, B: int):
    """Split the flat regression hist into (cells (G,Fp,B,2), node_s2 (G,))."""
    cells = flat[: G * Fp * B * 2].view(G, Fp, B, 2)
    return cells, flat[G * Fp * B * 2 :]


# ---------------------------------------------------------------------------
# K4b — node-assignment advance, all trees in one launch
# ---------------------------------------------------------------------------


def node_advance(
    codes: torch.Tensor,   # N×F uint8
    nodeid: torch.Tensor,  # T×N int32 GLOBAL arena ids (out-of-place result)
    feat: torch.Tensor,    # n_nodes int32: split feature per arena node (-1 leaf)
    thr: torch.Tensor,     # n_nodes int32: split bin (go left iff code <= thr)
    left: torch.Tensor,    # n_nodes int32 (global child ids)
    right: torch.Tensor,   # n_nodes int32
    catmask: Optional[torch.Tensor] = None,  # n_nodes int64: categorical
    # split-set (bit c set = category c left); 0 = continuous node
) -> torch.Tensor:
    """Advance every (tree,row) one level; rows at leaves keep their id (K4b)."""
    if catmask is None:
        catmask = torch.zeros_like(left, dtype=torch.int64)
    if hip_available(codes):
        return get_ext().node_advance(codes, nodeid, feat, thr, left, right,
                                      catmask)
    T, N = nodeid.shape
    out = torch.empty_like(nodeid)
    for t in range(T):
        out[t] = _node_advance_torch(codes, nodeid[t], feat, thr, left, right,
                                     catmask)
    return out


def _node_advance_torch(codes, nodeid, feat_of, thrbin_of, left_of, right_of,
                        catmask_of):
    nid = nodeid.to(torch.int64)
    live = nid >= 0
    nid_c = nid.clamp_min(0)
    f = feat_of.to(torch.int64)[nid_c]
    internal = (f >= 0) & live
    fx = codes.gather(1, f.clamp_min(0).unsqueeze(1)).squeeze(1).to(torch.int32)
    m = catmask_of[nid_c]
    go_left = torch.where(
        m != 0,
        (m >> fx.to(torch.int64).clamp_(0, 63)) & 1 == 1,
        fx <= thrbin_of[nid_c],
    )
    nxt = torch.where(go_left, left_of.to(torch.int64)[nid_c], right_of.to(torch.int64)[nid_c])
    # keep-at-leaf: rows at leaves retain their node id (histograms skip
    # them via lut = -1; the final per-leaf stats pass needs them live)
    out = torch.where(internal, nxt, nid)
    return out.to(torch.int32)


# ---------------------------------------------------------------------------
# K8 — batched ensemble tree inference
# ---------------------------------------------------------------------------


TREE_CAT_FLAG = 1 << 30


def pack_tree_nodes(feature: torch.Tensor, threshold: torch.Tensor,
                    left: torch.Tensor, right: torch.Tensor,
                    catmask: torch.Tensor) -> torch.Tensor:
    """T×M×4 int32 packed nodes for the K8 kernel: {feature | CAT_FLAG,
    threshold bits, left, right} — one 16-B load per visit instead of four
    scattered SoA reads."""
    f = feature.to(torch.int32).clone()
    f = torch.where((catmask != 0) & (f >= 0), f | TREE_CAT_FLAG, f)
    return torch.stack([
        f,
        threshold.to(torch.float32).view(torch.int32),
        left.to(torch.int32),
        right.to(torch.int32),
    ], dim=2).contiguous()


def tree_predict(
    X: torch.Tensor,         # N×F float32 raw features
    feature: torch.Tensor,   # T×M int32 (internal: feature id; leaf: -1)
    threshold: torch.Tensor, # T×M float32
    left: torch.Tensor,      # T×M int32
    right: torch.Tensor,     # T×M int32
    value: torch.Tensor,     # T×M×K float32 leaf vectors
    catmask: Optional[torch.Tensor] = None,  # T×M int64 split-sets (0 = cont)
    packed: Optional[torch.Tensor] = None,   # prebuilt pack_tree_nodes output
) -> torch.Tensor:
    """Sum of per-tree leaf value vectors per row: N×K float32 (K8).

    Categorical nodes route by set membership: category = int(x), go left
    iff bit ``category`` of ``catmask`` is set (CategoricalSplit.shouldGoLeft;
    unseen categories go right, as in Spark)."""
    if catmask is None:
        catmask = torch.zeros_like(left, dtype=torch.int64)
    if hip_available(X):
        if packed is None:
            packed = pack_tree_nodes(feature, threshold, left, right, catmask)
        return get_ext().tree_predict(
            X.contiguous(), packed.contiguous(), value.contiguous(),
            catmask.contiguous()
        )
    return _tree_predict_torch(X, feature, threshold, left, right, value, catmask)


def _tree_predict_torch(X, feature, threshold, left, right, value, catmask):
    N, F = X.shape
    T, M = feature.shape
    K = value.shape[2]
    out = torch.zeros(N, K, dtype=torch.float32, device=X.device)
    for t in range(T):
        node = torch.zeros(N, dtype=torch.int64, device=X.device)
        feat_t = feature[t].to(torch.int64)
        thr_t = threshold[t]
        left_t = left[t].to(torch.int64)
        right_t = right[t].to(torch.int64)
        mask_t = catmask[t]
        for _ in range(64):
            f = feat_t[node]
            internal = f >= 0
            if not bool(internal.any()):
                break
            fx = X.gather(1, torch.clamp(f, min=0).unsqueeze(1)).squeeze(1)
            m = mask_t[node]
            cat = fx.to(torch.int64).clamp(0, 63)
            go_left = torch.where(
                m != 0, ((m >> cat) & 1 == 1) & (fx >= 0), fx <= thr_t[node]
            )
            nxt = torch.where(go_left, left_t[node], right_t[node])
            node = torch.where(internal, nxt, node)
        out += value[t][node]
    return out


# ---------------------------------------------------------------------------
# K1 — fused multinomial logistic gradient
# ---------------------------------------------------------------------------


def logreg_grad(
    X: torch.Tensor,  # N×F float32
    y: torch.Tensor,  # N int
    w: Optional[torch.Tensor],  # N float32 or None
    W: torch.Tensor,  # K×F float32
    b: torch.Tensor,  # K float32
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Partial (unnormalized) multinomial logloss + gradient over the shard.

    Returns (loss_sum scalar, gradW K×F, gradB K) fp32 — caller all-reduces
    then normalizes. Reference site: LogisticAggregator.add
    inside treeAggregate.
    """
    if hip_available(X):
        # Composition of hand kernels: logits via the coalesced f32 MFMA
        # GEMM (K5) + fused softmax/xent/delta (K6) — ~10x the throughput of
        # the per-thread-dot variant (X rows are only 164 B, so per-lane
        # sequential reads split 41 ways per wave instruction). gradW = deltaᵀX
        # is a plain GEMM -> rocBLAS (sanctioned for plain library GEMMs).
        logits = gemm_bias_act(X, W, b, "none")
        loss, delta = get_ext().softmax_xent_grad(
            logits.contiguous(), y.to(torch.int32).contiguous(),
            w.contiguous() if w is not None else _empty(X.device),
        )
        # gradW = deltaᵀX is a C×F×N outer product; rocBLAS picks a
        # no-split-K tile for it (2 workgroups on 256 CUs, ~2 ms) — the
        # split-K kernel writes exact per-block partials and sums them in
        # fixed order (deterministic), with the bias grad fused.
        if delta.size(1) * (X.size(1) + 1) <= 1024:
            gradW, gradB = get_ext().wgrad_f32(X, delta.contiguous())
        else:
            gradW = delta.T @ X
            gradB = delta.sum(dim=0)
        return loss, gradW, gradB
    return _logreg_grad_torch(X, y, w, W, b)


def _logreg_grad_torch(X, y, w, W, b):
    logits = X @ W.T + b
    logp = torch.log_softmax(logits.to(torch.float32), dim=1)
    yl = y.to(torch.int64)
    nll = -logp.gather(1, yl.unsqueeze(1)).squeeze(1)
    delta = logp.exp().clone()
    delta.scatter_add_(1, yl.unsqueeze(1), -torch.ones_like(nll).unsqueeze(1))
    if w is not None and w.numel():
        loss = (nll * w).sum()
        delta = delta * w.unsqueeze(1)
    else:
        loss = nll.sum()
    gradW = delta.T @ X
    gradB = delta.sum(dim=0)
    return loss.to(torch.float32), gradW.to(torch.float32), gradB.to(torch.float32)


# ---------------------------------------------------------------------------
# K5-K7 — MLP fused forward/backward (GEMM+bias+activation, softmax-xent)
# ---------------------------------------------------------------------------


def gemm_bias_act(
    X: torch.Tensor,  # N×K in
    W: torch.Tensor,  # M×K (out×in)
    b: torch.Tensor,  # M
    act: str = "none",
) -> torch.Tensor:
    """out = act(X @ W.T + b) — the K5 fused MFMA GEMM on GPU."""
    if hip_available(X):
        return get_ext().gemm_bias_act(
            X.contiguous(), W.contiguous(), b.contiguous(), _act_code(act)
        )
    h = X @ W.T + b
    return _act(h, act)


def _act(h: torch.Tensor, activation: str) -> torch.Tensor:
    if activation in ("none", None):
        return h
    if activation == "relu":
        return torch.relu(h)
    if activation == "sigmoid":
        return torch.sigmoid(h)
    raise ValueError(f"Unsupported activation {activation!r}")


def _act_code(activation: str) -> int:
    return {"none": 0, "relu": 1, "sigmoid": 2}[activation]


def mlp_forward(
    X: torch.Tensor,
    Ws: List[torch.Tensor],
    bs: List[torch.Tensor],
    activation: str = "relu",
    out_dtype: torch.dtype = torch.float32,
) -> torch.Tensor:
    """Forward pass: N×K logits (no softmax). Reference site: BLAS.gemm in
    spark ml/ann AffineLayer."""
    h = X
    L = len(Ws)
    for i, (W, b) in enumerate(zip(Ws, bs)):
        h = gemm_bias_act(h, W, b, "none" if i == L - 1 else activation)
    return h.to(out_dtype)


def mlp_grad(
    X: torch.Tensor,
    y: torch.Tensor,
    Ws: List[torch.Tensor],
    bs: List[torch.Tensor],
    activation: str = "relu",
    w: Optional[torch.Tensor] = None,
    in_features: Optional[int] = None,
):
    """Fused forward+backward for the softmax/cross-entropy MLP.

    Returns (loss_sum fp32, [gradW...], [gradB...]) — unnormalized partial
    sums over the local shard (caller all-reduces: SURVEY.md §2.5 C3).
    Reference sites: ANNGradient.compute / BreezeUtil.dgemm (K5-K7).
    """
    L = len(Ws)
    acts = [X]
    h = X
    for i in range(L):
        h = gemm_bias_act(h, Ws[i], bs[i], "none" if i == L - 1 else activation)
        acts.append(h)
    logits = acts[-1].to(torch.float32)
    loss, delta = softmax_xent_grad(logits, y, w)
    delta = delta.to(X.dtype)
    gWs: List[Optional[torch.Tensor]] = [None] * L
    gBs: List[Optional[torch.Tensor]] = [None] * L
    use_wgrad_kernel = (
        X.is_cuda and X.dtype == torch.bfloat16 and hip_available(X)
    )
    for i in range(L - 1, -1, -1):
        a = acts[i]
        if use_wgrad_kernel:
            # K7 split-N MFMA wgrad with fused bias grad (rocBLAS picks a
            # 2-workgroup no-split-K tile for these K=1M outer products)
            gWs[i], gBs[i] = get_ext().wgrad_bias(delta.contiguous(), a.contiguous())
        else:
            gWs[i] = matmul_tn(delta, a)      # K7 wgrad: (N×K)ᵀ·(N×F)
            gBs[i] = delta.sum(dim=0).to(torch.float32)
        if i == 0 and in_features is not None and gWs[0].shape[1] != in_features:
            gWs[0] = gWs[0][:, :in_features].contiguous()  # drop X pad columns
        if i > 0:
            h_i = acts[i]
            out_i, in_i = Ws[i].shape
            if (
                delta.is_cuda and delta.dtype == torch.bfloat16
                and h_i.dtype == torch.bfloat16 and in_i <= 128
                and hip_available(delta)
            ):
                # fused dgrad+act': one MFMA kernel writes the next delta
                # directly (was rocBLAS dgrad + a separate act_backward
                # read-modify-write — ~0.28 ms of the bf16 step)
                Kp = ((out_i + 31) // 32) * 32
                d_in = delta
                Wt = Ws[i].T
                if Kp != out_i:
                    d_in = torch.nn.functional.pad(delta, (0, Kp - out_i))
                    Wt = torch.nn.functional.pad(Wt, (0, Kp - out_i))
                delta = get_ext().gemm_dgrad_act(
                    d_in.contiguous(), Wt.contiguous(), h_i.contiguous(),
                    _act_code(activation))
            else:
                dx = delta @ Ws[i]            # K7 dgrad (plain GEMM)
                if activation == "relu":
                    dact = (h_i > 0).to(dx.dtype)
                else:
                    dact = h_i * (1 - h_i)
                delta = dx * dact
    return loss, gWs, gBs


def softmax_xent_grad(
    logits: torch.Tensor, y: torch.Tensor, w: Optional[torch.Tensor] = None
) -> Tuple[torch.Tensor, torch.Tensor]:
    """(loss_sum, delta = softmax - onehot), weighted — K6."""
    if hip_available(logits):
        return get_ext().softmax_xent_grad(
            logits.contiguous(), y.to(torch.int32).contiguous(),
            w.contiguous() if w is not None else _empty(logits.device),
        )
    logp = torch.log_softmax(logits.to(torch.float32), dim=1)
    yl = y.to(torch.int64)
    nll = -logp.gather(1, yl.unsqueeze(1)).squeeze(1)
    delta = logp.exp().clone()
    delta.scatter_add_(1, yl.unsqueeze(1), -torch.ones_like(nll).unsqueeze(1))
    if w is not None and w.numel():
        loss = (nll * w).sum()
        delta = delta * w.unsqueeze(1)
    else:
        loss = nll.sum()
    return loss.to(torch.float32), delta


def matmul_tn(A: torch.Tensor, B: torch.Tensor) -> torch.Tensor:
    """wgrad helper: ``A.T @ B`` as fp32, A: N×K, B: N×F → K×F.

    This is a *plain* GEMM (no fusion), so rocBLAS via torch.matmul is the
    sanctioned path on GPU (north star: "hipBLASLt/rocBLAS only for plain
    library GEMMs"). bf16 operands stay bf16 through rocBLAS (fp32 internal
    accumulation) — the fp32 up-cast of a 1M-row activation would cost more
    HBM than the GEMM itself; only the small K×F result is up-cast.
    """
    if A.is_cuda and A.dtype == torch.bfloat16 and B.dtype == torch.bfloat16:
        return (A.T @ B).to(torch.float32)
    return A.T.to(torch.float32) @ B.to(torch.float32)


def split_scan_packed(
    hist: torch.Tensor,  # G×F'×B×S float32 (already all-reduced)
    impurity: str,
    min_instances: int,
    min_info_gain: float,
) -> torch.Tensor:
    """K3 — best split per node, packed per-node rows
    [best_f, best_bin, gain, imp_l, imp_r, cnt_l, cnt_r, ls[S], rs[S],
    imp_self, parent[S]] (G×(8+3S) float32). The trailing node-own
    impurity/stats let the grower skip a dedicated root-stats pass. GPU: one
    workgroup per node, LDS-staged slab, double gain math in the torch
    reference's exact formula order (bitwise match).
    """
    code = {"gini": 0, "entropy": 1, "variance": 2}[impurity]
    if hip_available(hist):
        return get_ext().split_scan(
            hist.contiguous(), code, int(min_instances), float(min_info_gain)
        )
    from flowml.classification.tree import (
        _counts_of,
        _impurity_from_stats,
        split_scan,
    )

    G, Fp, B, S = hist.shape
    best_f, best_bin, best_gain, parent, left_stats = split_scan(
        hist, impurity, min_instances, min_info_gain
    )
    right_stats = parent - left_stats
    ok = best_f >= 0
    z = torch.zeros_like
    left_stats = torch.where(ok.unsqueeze(1), left_stats, z(left_stats))
    right_stats = torch.where(ok.unsqueeze(1), right_stats, z(right_stats))
    imp_l = torch.where(ok, _impurity_from_stats(left_stats, impurity), z(best_gain))
    imp_r = torch.where(ok, _impurity_from_stats(right_stats, impurity), z(best_gain))
    cnt_l = torch.where(ok, _counts_of(left_stats, impurity), z(best_gain))
    cnt_r = torch.where(ok, _counts_of(right_stats, impurity), z(best_gain))
    imp_p = _impurity_from_stats(parent, impurity)
    return torch.cat(
        [
            best_f.to(torch.float64).reshape(G, 1),
            torch.where(ok, best_bin, torch.zeros_like(best_bin)).to(torch.float64).reshape(G, 1),
            best_gain.reshape(G, 1),
            imp_l.reshape(G, 1),
            imp_r.reshape(G, 1),
            cnt_l.reshape(G, 1),
            cnt_r.reshape(G, 1),
            left_stats,
            right_stats,
            imp_p.reshape(G, 1),
            parent,
        ],
        dim=1,
    ).to(torch.float32)
