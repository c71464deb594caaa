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
"""

from __future__ import annotations

import os
from typing import Optional, Tuple

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


# ---------------------------------------------------------------------------
# K1 — fused multinomial logistic gradient
# ---------------------------------------------------------------------------


def logreg_grad(
    X: torch.Tensor,  # N×F float32
    y: torch.Tensor,  # N int32/int64
    w: Optional[torch.Tensor],  # N float32 or None
    W: torch.Tensor,  # K×F float32 (coefficients)
    b: torch.Tensor,  # K float32
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Partial (unnormalized) multinomial logloss + gradient over the shard.

    Returns (loss_sum scalar, gradW K×F, gradB K) — all fp32, NOT divided by
    the weight sum (the caller all-reduces then normalizes).
    Reference site: LogisticAggregator.add inside treeAggregate.
    """
    if hip_available(X):
        ext = get_ext()
        return ext.logreg_grad(X.contiguous(), y.to(torch.int32).contiguous(),
                               w.contiguous() if w is not None else torch.empty(0, device=X.device),
                               W.contiguous(), b.contiguous())
    return _logreg_grad_torch(X, y, w, W, b)


def _logreg_grad_torch(X, y, w, W, b):
    logits = X @ W.T + b  # N×K
    logp = torch.log_softmax(logits.to(torch.float32), dim=1)
    yl = y.to(torch.int64)
    nll = -logp.gather(1, yl.unsqueeze(1)).squeeze(1)
    p = logp.exp()
    delta = p  # N×K
    delta = delta.clone()
    delta.scatter_add_(1, yl.unsqueeze(1), -torch.ones_like(nll).unsqueeze(1))
    if w is not None and w.numel():
        loss = (nll * w).sum()
        delta = delta * w.unsqueeze(1)
    else:
        loss = nll.sum()
    gradW = delta.T @ X  # K×F
    gradB = delta.sum(dim=0)
    return loss.to(torch.float32), gradW.to(torch.float32), gradB.to(torch.float32)


# ---------------------------------------------------------------------------
# K4a — quantile binning: codes = searchsorted(thresholds, x)
# ---------------------------------------------------------------------------


def bin_codes(X: torch.Tensor, thresholds: torch.Tensor) -> torch.Tensor:
    """Quantize N×F float features to uint8 bin codes given per-feature
    threshold rows (F×(B-1), padded with +inf). code = #thresholds <= x.

    Reference site: TreePoint.convertToTreeRDD.
    """
    if hip_available(X):
        return get_ext().bin_codes(X.contiguous(), thresholds.contiguous())
    return _bin_codes_torch(X, thresholds)


def _bin_codes_torch(X, thresholds):
    # thresholds: F×T ascending (+inf padded); code_f(x) = sum(t <= x)
    codes = torch.searchsorted(
        thresholds.to(X.dtype), X.T.contiguous(), right=True
    ).T  # N×F
    return codes.to(torch.uint8)


# ---------------------------------------------------------------------------
# K2 — per-node class histograms over binned features
# ---------------------------------------------------------------------------


def tree_hist(
    codes: torch.Tensor,  # N×F uint8 bin codes
    node_idx: torch.Tensor,  # N int32: index into the node group, -1 = not in group
    labels: torch.Tensor,  # N uint8/int
    weights: Optional[torch.Tensor],  # N float32 Poisson bag counts (or None)
    n_nodes: int,
    n_bins: int,
    n_classes: int,
    feat_subset: Optional[torch.Tensor] = None,  # G×F' int32 per-node feature ids
) -> torch.Tensor:
    """Weighted class histogram: out[g, f', b, c] += w_i.

    Returns float32 tensor of shape (n_nodes, F', n_bins, n_classes), where
    F' = codes.shape[1] when feat_subset is None else feat_subset.shape[1].
    Reference site: DTStatsAggregator bin update (SURVEY.md §2.4 K2 — the #1
    hot loop of the flagship config).
    """
    if hip_available(codes):
        ext = get_ext()
        w = weights if weights is not None else torch.empty(0, device=codes.device)
        fs = feat_subset if feat_subset is not None else torch.empty(0, dtype=torch.int32, device=codes.device)
        return ext.tree_hist(codes, node_idx.to(torch.int32), labels.to(torch.uint8),
                             w, int(n_nodes), int(n_bins), int(n_classes), fs)
    return _tree_hist_torch(codes, node_idx, labels, weights, n_nodes, n_bins, n_classes, feat_subset)


def _tree_hist_torch(codes, node_idx, labels, weights, n_nodes, n_bins, n_classes, feat_subset):
    N, F = codes.shape
    dev = codes.device
    node_idx = node_idx.to(torch.int64)
    active = node_idx >= 0
    codes_a = codes[active].to(torch.int64)
    nodes_a = node_idx[active]
    labels_a = labels[active].to(torch.int64)
    w_a = (weights[active].to(torch.float64) if weights is not None and weights.numel()
           else torch.ones(len(nodes_a), dtype=torch.float64, device=dev))
    if feat_subset is not None and feat_subset.numel():
        Fp = feat_subset.shape[1]
        fs = feat_subset.to(torch.int64)  # G×F'
        # per active row: gather its node's feature subset
        feats = fs[nodes_a]  # n_a × F'
        c = codes_a.gather(1, feats)  # n_a × F'
    else:
        Fp = F
        c = codes_a
    # flat index: ((g*Fp + f)*n_bins + bin)*n_classes + label
    g = nodes_a.unsqueeze(1)
    f = torch.arange(Fp, dtype=torch.int64, device=dev).unsqueeze(0)
    lab = labels_a.unsqueeze(1)
    flat = ((g * Fp + f) * n_bins + c) * n_classes + lab
    out = torch.zeros(n_nodes * Fp * n_bins * n_classes, dtype=torch.float64, device=dev)
    out.scatter_add_(0, flat.reshape(-1), w_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
    return out.reshape(n_nodes, Fp, n_bins, n_classes).to(torch.float32)


# ---------------------------------------------------------------------------
# K2b — per-node (grad, hess, count) histograms for GBT regression trees
# ---------------------------------------------------------------------------


def tree_hist_grad(
    codes: torch.Tensor,
    node_idx: torch.Tensor,
    grad: torch.Tensor,  # N float32
    hess: torch.Tensor,  # N float32
    weights: Optional[torch.Tensor],
    n_nodes: int,
    n_bins: int,
    feat_subset: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """out[g, f, b, 0..2] += (w*grad, w*hess, w). Shape (G,F',B,3) float32."""
    if hip_available(codes):
        ext = get_ext()
        w = weights if weights is not None else torch.empty(0, device=codes.device)
        fs = feat_subset if feat_subset is not None else torch.empty(0, dtype=torch.int32, device=codes.device)
        return ext.tree_hist_grad(codes, node_idx.to(torch.int32), grad, hess, w,
                                  int(n_nodes), int(n_bins), fs)
    return _tree_hist_grad_torch(codes, node_idx, grad, hess, weights, n_nodes, n_bins, feat_subset)


def _tree_hist_grad_torch(codes, node_idx, grad, hess, weights, n_nodes, n_bins, feat_subset):
    N, F = codes.shape
    dev = codes.device
    node_idx = node_idx.to(torch.int64)
    active = node_idx >= 0
    codes_a = codes[active].to(torch.int64)
    nodes_a = node_idx[active]
    g_a = grad[active].to(torch.float64)
    h_a = hess[active].to(torch.float64)
    w_a = (weights[active].to(torch.float64) if weights is not None and weights.numel()
           else torch.ones(len(nodes_a), dtype=torch.float64, device=dev))
    g_a = g_a * w_a
    h_a = h_a * w_a
    if feat_subset is not None and feat_subset.numel():
        Fp = feat_subset.shape[1]
        feats = feat_subset.to(torch.int64)[nodes_a]
        c = codes_a.gather(1, feats)
    else:
        Fp = F
        c = codes_a
    gg = nodes_a.unsqueeze(1)
    f = torch.arange(Fp, dtype=torch.int64, device=dev).unsqueeze(0)
    flat = ((gg * Fp + f) * n_bins + c) * 3
    out = torch.zeros(n_nodes * Fp * n_bins * 3, dtype=torch.float64, device=dev)
    out.scatter_add_(0, flat.reshape(-1), g_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
    out.scatter_add_(0, (flat + 1).reshape(-1), h_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
    out.scatter_add_(0, (flat + 2).reshape(-1), w_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
    return out.reshape(n_nodes, Fp, n_bins, 3).to(torch.float32)


# ---------------------------------------------------------------------------
# K8 — batched ensemble tree inference
# ---------------------------------------------------------------------------


def tree_predict(
    X: torch.Tensor,  # N×F float32 raw features
    feature: torch.Tensor,  # T×M int32 (internal: feature id; leaf: -1)
    threshold: torch.Tensor,  # T×M float32
    left: torch.Tensor,  # T×M int32 child index (node-local)
    right: torch.Tensor,  # T×M int32
    value: torch.Tensor,  # T×M×K float32 leaf class scores
) -> torch.Tensor:
    """Sum of per-tree leaf value vectors per row: N×K float32.

    Breadth-first node arrays per tree, padded to max node count M.
    Reference site: predictRaw UDF tree traversal.
    """
    if hip_available(X):
        return get_ext().tree_predict(X.contiguous(), feature, threshold, left, right, value)
    return _tree_predict_torch(X, feature, threshold, left, right, value)


def _tree_predict_torch(X, feature, threshold, left, right, value):
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
        # max depth bounded by M
        for _ in range(64):
            f = feat_t[node]
            internal = f >= 0
            if not bool(internal.any()):
                break
            fx = X.gather(1, torch.clamp(f, min=0).unsqueeze(1)).squeeze(1)
            go_left = fx <= thr_t[node]
            nxt = torch.where(go_left, left_t[node], right_t[node])
            node = torch.where(internal, nxt, node)
        out += value[t][node]
    return out


# ---------------------------------------------------------------------------
# K5-K7 — MLP fused forward/backward (GEMM+bias+activation, softmax-xent)
# ---------------------------------------------------------------------------


def mlp_forward(
    X: torch.Tensor,
    Ws: "list[torch.Tensor]",
    bs: "list[torch.Tensor]",
    activation: str = "relu",
    out_dtype: torch.dtype = torch.float32,
) -> torch.Tensor:
    """Forward pass: returns N×K logits (no softmax).

    Reference site: BLAS.gemm in spark ml/ann AffineLayer.
    GPU path: fused MFMA GEMM+bias+ReLU HIP kernel per layer.
    """
    if hip_available(X):
        ext = get_ext()
        h = X
        for i, (W, b) in enumerate(zip(Ws, bs)):
            last = i == len(Ws) - 1
            h = ext.gemm_bias_act(h, W, b, 0 if last else _act_code(activation))
        return h.to(out_dtype)
    h = X
    for i, (W, b) in enumerate(zip(Ws, bs)):
        h = h @ W.T + b
        if i < len(Ws) - 1:
            h = _act(h, activation)
    return h.to(out_dtype)


def _act(h: torch.Tensor, activation: str) -> torch.Tensor:
    if activation == "relu":
        return torch.relu(h)
    if activation == "sigmoid":
        return torch.sigmoid(h)
    raise ValueError(f"Unsupported activation {activation!r}")


def _act_code(activation: str) -> int:
    return {"relu": 1, "sigmoid": 2}[activation]


def mlp_grad(
    X: torch.Tensor,  # N×F
    y: torch.Tensor,  # N int64
    Ws: "list[torch.Tensor]",
    bs: "list[torch.Tensor]",
    activation: str = "relu",
    w: "Optional[torch.Tensor]" = None,
):
    """Fused forward+backward for the softmax/cross-entropy MLP.

    Returns (loss_sum fp32 scalar, [gradW...], [gradB...]) — unnormalized
    partial sums over the local shard (caller all-reduces: SURVEY.md §2.5 C3).
    Reference sites: ANNGradient.compute / BreezeUtil.dgemm (K5-K7).
    """
    # forward, keeping activations
    acts = [X]
    h = X
    L = len(Ws)
    for i in range(L):
        z = h @ Ws[i].T + bs[i]
        h = _act(z, activation) if i < L - 1 else z
        acts.append(h)
    logits = acts[-1].to(torch.float32)
    logp = torch.log_softmax(logits, dim=1)
    yl = y.to(torch.int64)
    nll = -logp.gather(1, yl.unsqueeze(1)).squeeze(1)
    delta = logp.exp()
    delta.scatter_add_(1, yl.unsqueeze(1), -torch.ones_like(nll).unsqueeze(1))
    if w is not None and w.numel():
        loss = (nll * w).sum()
        delta = delta * w.unsqueeze(1)
    else:
        loss = nll.sum()
    delta = delta.to(X.dtype)
    gWs = [None] * L
    gBs = [None] * L
    for i in range(L - 1, -1, -1):
        a = acts[i]
        gWs[i] = (delta.T @ a).to(torch.float32)  # K7 wgrad
        gBs[i] = delta.sum(dim=0).to(torch.float32)
        if i > 0:
            dx = delta @ Ws[i]  # K7 dgrad
            h_i = acts[i]
            if activation == "relu":
                dact = (h_i > 0).to(dx.dtype)
            else:  # sigmoid
                dact = h_i * (1 - h_i)
            delta = dx * dact
    return loss.to(torch.float32), gWs, gBs


# ---------------------------------------------------------------------------
# K2 (multi-tree form) — the flagship hot loop: all trees' node groups in one
# pass over the binned shard (SURVEY.md §2.3 "one histogram pass serves all
# trees in the node group")
# ---------------------------------------------------------------------------


def tree_hist_multi(
    codes: torch.Tensor,      # N×F uint8
    group_idx: torch.Tensor,  # N×T int32: frontier-group index per (row,tree), -1 inactive
    labels: torch.Tensor,     # N int (class) — or empty for regression
    bag: Optional[torch.Tensor],  # N×T uint8/float Poisson bag counts (None = 1)
    n_groups: int,
    n_bins: int,
    n_classes: int,
    feat_subset: Optional[torch.Tensor] = None,  # G×F' int32
    targets: Optional[torch.Tensor] = None,  # N float32 (regression: per-row target)
) -> torch.Tensor:
    """Classification: out[g,f',b,c] += bag[i,t]  (shape G×F'×B×C).
    Regression (targets given): out[g,f',b,:] += bag*(t, t², 1) (shape G×F'×B×3).
    """
    if hip_available(codes):
        ext = get_ext()
        fs = feat_subset if feat_subset is not None else torch.empty(0, dtype=torch.int32, device=codes.device)
        bg = bag if bag is not None else torch.empty(0, dtype=torch.uint8, device=codes.device)
        if targets is None:
            return ext.tree_hist_multi(codes, group_idx.to(torch.int32), labels.to(torch.uint8),
                                       bg, int(n_groups), int(n_bins), int(n_classes), fs)
        return ext.tree_hist_multi_reg(codes, group_idx.to(torch.int32), targets, bg,
                                       int(n_groups), int(n_bins), fs)
    return _tree_hist_multi_torch(codes, group_idx, labels, bag, n_groups, n_bins,
                                  n_classes, feat_subset, targets)


def _tree_hist_multi_torch(codes, group_idx, labels, bag, n_groups, n_bins,
                           n_classes, feat_subset, targets):
    N, F = codes.shape
    T = group_idx.shape[1]
    dev = codes.device
    gi = group_idx.to(torch.int64)
    stat_w = 3 if targets is not None else n_classes
    Fp = feat_subset.shape[1] if (feat_subset is not None and feat_subset.numel()) else F
    out = torch.zeros(n_groups * Fp * n_bins * stat_w, dtype=torch.float64, device=dev)
    for t in range(T):
        g_t = gi[:, t]
        active = g_t >= 0
        if bag is not None:
            w_t = bag[:, t].to(torch.float64)
            active = active & (w_t > 0)
        if not bool(active.any()):
            continue
        rows = torch.nonzero(active, as_tuple=False).flatten()
        g_a = g_t[rows]
        c_a = codes[rows].to(torch.int64)
        w_a = (bag[rows, t].to(torch.float64) if bag is not None
               else torch.ones(len(rows), dtype=torch.float64, device=dev))
        if feat_subset is not None and feat_subset.numel():
            feats = feat_subset.to(torch.int64)[g_a]  # n_a×F'
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
                             (w_a * tv * tv).unsqueeze(1).expand(-1, Fp).reshape(-1))
            out.scatter_add_(0, (base + 2).reshape(-1),
                             w_a.unsqueeze(1).expand(-1, Fp).reshape(-1))
    return out.reshape(n_groups, Fp, n_bins, stat_w).to(torch.float32)


def node_advance(
    codes: torch.Tensor,      # N×F uint8
    nodeid: torch.Tensor,     # N int32 current node per row (one tree), -1 done
    feat_of: torch.Tensor,    # M int32 (split feature per node, -1 leaf)
    thrbin_of: torch.Tensor,  # M int32 (split bin: go left iff code <= thr)
    left_of: torch.Tensor,    # M int32
    right_of: torch.Tensor,   # M int32
) -> torch.Tensor:
    """Advance each row one level down its tree (K4b node-id update)."""
    if hip_available(codes):
        return get_ext().node_advance(codes, nodeid.to(torch.int32), feat_of, thrbin_of, left_of, right_of)
    nid = nodeid.to(torch.int64)
    live = nid >= 0
    nid_c = torch.clamp(nid, min=0)
    f = feat_of.to(torch.int64)[nid_c]
    internal = (f >= 0) & live
    fx = codes.gather(1, torch.clamp(f, min=0).unsqueeze(1)).squeeze(1).to(torch.int32)
    go_left = fx <= thrbin_of[nid_c]
    nxt = torch.where(go_left, left_of.to(torch.int64)[nid_c], right_of.to(torch.int64)[nid_c])
    out = torch.where(internal, nxt, torch.full_like(nid, -1))
    out = torch.where(live, out, nid)
    return out.to(torch.int32)
