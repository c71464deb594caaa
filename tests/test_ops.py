"""Ops reference implementations vs independent oracles (SURVEY.md §4
"kernel parity" — these CPU references are what the HIP kernels are later
tested against, so they get their own naive-python oracles here)."""

import numpy as np
import pytest
import torch

from flowml import ops


class TestLogregGrad:
    def test_matches_autograd(self):
        torch.manual_seed(0)
        N, F, K = 64, 7, 4
        X = torch.randn(N, F)
        y = torch.randint(0, K, (N,))
        W = torch.randn(K, F, requires_grad=True)
        b = torch.randn(K, requires_grad=True)
        loss, gW, gB = ops.logreg_grad(X, y, None, W.detach(), b.detach())
        ref = torch.nn.functional.cross_entropy(X @ W.T + b, y, reduction="sum")
        ref.backward()
        assert float(loss) == pytest.approx(float(ref.detach()), rel=1e-5)
        assert torch.allclose(gW, W.grad, atol=1e-4)
        assert torch.allclose(gB, b.grad, atol=1e-4)

    def test_weighted(self):
        torch.manual_seed(1)
        N, F, K = 32, 5, 3
        X = torch.randn(N, F)
        y = torch.randint(0, K, (N,))
        w = torch.rand(N) + 0.5
        W = torch.randn(K, F, requires_grad=True)
        b = torch.zeros(K, requires_grad=True)
        loss, gW, gB = ops.logreg_grad(X, y, w, W.detach(), b.detach())
        ref = (torch.nn.functional.cross_entropy(X @ W.T + b, y, reduction="none") * w).sum()
        ref.backward()
        assert float(loss) == pytest.approx(float(ref.detach()), rel=1e-5)
        assert torch.allclose(gW, W.grad, atol=1e-4)


class TestBinCodes:
    def test_against_naive(self):
        r = np.random.default_rng(0)
        X = torch.from_numpy(r.normal(size=(100, 3)).astype(np.float32))
        thr = torch.sort(torch.from_numpy(r.normal(size=(3, 7)).astype(np.float32)), dim=1).values
        codes = ops.bin_codes(X, thr)
        for i in range(100):
            for f in range(3):
                expect = int((thr[f] < X[i, f]).sum())  # strict: code<=b ⟺ x<=thr[b]
                assert int(codes[i, f]) == expect

    def test_inf_padding(self):
        X = torch.tensor([[5.0]])
        thr = torch.tensor([[1.0, 2.0, float("inf"), float("inf")]])
        assert int(ops.bin_codes(X, thr)[0, 0]) == 2


def naive_hist(codes, nodeid, lut, labels, bag, G, B, C, fs, targets=None):
    """Independent python-loop oracle for K2 (classification & regression).
    nodeid holds GLOBAL arena ids; lut is 1-D arena id -> group."""
    N, F = codes.shape
    T = nodeid.shape[0]
    Fp = fs.shape[1] if fs is not None else F
    S = 2 if targets is not None else C
    out = np.zeros((G, Fp, B, S), dtype=np.float64)
    ns2 = np.zeros(G, dtype=np.float64)
    for t in range(T):
        for i in range(N):
            nid = int(nodeid[t, i])
            if nid < 0:
                continue
            g = int(lut[nid])
            if g < 0:
                continue
            w = float(bag[t, i]) if bag is not None else 1.0
            if w == 0:
                continue
            if targets is not None:
                ns2[g] += w * float(targets[i]) * float(targets[i])
            for fp in range(Fp):
                f = int(fs[g, fp]) if fs is not None else fp
                b = int(codes[i, f])
                if targets is None:
                    out[g, fp, b, int(labels[i])] += w
                else:
                    tv = float(targets[i])
                    out[g, fp, b, 0] += w * tv
                    out[g, fp, b, 1] += w
    if targets is not None:
        return out, ns2
    return out


class TestTreeHistNodes:
    def test_bitwise_vs_naive(self):
        r = np.random.default_rng(3)
        N, F, T, B, C = 80, 5, 3, 8, 3
        M = 4  # nodes per tree (global arena id = t*M + local)
        codes = torch.from_numpy(r.integers(0, B, (N, F)).astype(np.uint8))
        local = r.integers(-1, M, (T, N)).astype(np.int32)
        glob = np.where(local >= 0, local + (np.arange(T)[:, None] * M), -1)
        nodeid = torch.from_numpy(glob.astype(np.int32))
        # groups: tree-major contiguous, 2 active nodes per tree; 1-D lut
        lut = torch.full((T * M,), -1, dtype=torch.int32)
        g = 0
        tree_lo = torch.zeros(T, dtype=torch.int32)
        for t in range(T):
            tree_lo[t] = g
            for nid in sorted(r.choice(M, 2, replace=False)):
                lut[t * M + nid] = g
                g += 1
        G = g
        labels = torch.from_numpy(r.integers(0, C, N).astype(np.uint8))
        bag = torch.from_numpy(r.integers(0, 3, (T, N)).astype(np.float32))
        fs = torch.from_numpy(
            np.stack([np.sort(r.choice(F, 3, replace=False)) for _ in range(G)]).astype(np.int32)
        )
        tree_cnt = torch.full((T,), 2, dtype=torch.int32)
        out = ops.tree_hist_nodes(codes, nodeid, lut, tree_lo, tree_cnt, labels, bag, G, 2, B, C, fs)
        ref = naive_hist(codes.numpy(), nodeid.numpy(), lut.numpy(), labels.numpy(),
                         bag.numpy(), G, B, C, fs.numpy())
        # integer counts must match exactly
        assert np.array_equal(out.numpy(), ref.astype(np.float32))

    def test_no_bag_no_subset(self):
        r = np.random.default_rng(4)
        N, F, B, C = 50, 4, 6, 2
        codes = torch.from_numpy(r.integers(0, B, (N, F)).astype(np.uint8))
        nodeid = torch.zeros((1, N), dtype=torch.int32)
        lut = torch.zeros(1, dtype=torch.int32)
        tree_lo = torch.zeros(1, dtype=torch.int32)
        labels = torch.from_numpy(r.integers(0, C, N).astype(np.uint8))
        tree_cnt = torch.ones(1, dtype=torch.int32)
        out = ops.tree_hist_nodes(codes, nodeid, lut, tree_lo, tree_cnt, labels, None, 1, 1, B, C, None)
        assert float(out.sum()) == N * F
        ref = naive_hist(codes.numpy(), nodeid.numpy(), lut.numpy(), labels.numpy(),
                         None, 1, B, C, None)
        assert np.array_equal(out.numpy(), ref.astype(np.float32))

    def test_regression_stats(self):
        r = np.random.default_rng(5)
        N, F, B = 40, 3, 4
        codes = torch.from_numpy(r.integers(0, B, (N, F)).astype(np.uint8))
        nodeid = torch.zeros((1, N), dtype=torch.int32)
        lut = torch.zeros(1, dtype=torch.int32)
        tree_lo = torch.zeros(1, dtype=torch.int32)
        targets = torch.from_numpy(r.normal(size=N).astype(np.float32))
        labels = torch.zeros(N, dtype=torch.uint8)
        tree_cnt = torch.ones(1, dtype=torch.int32)
        flat = ops.tree_hist_nodes(codes, nodeid, lut, tree_lo, tree_cnt, labels, None, 1, 1, B, 0,
                                   None, targets=targets)
        out, ns2 = ops.reg_hist_views(flat, 1, F, B)
        assert out.shape == (1, F, B, 2)
        for f in range(F):
            assert float(out[0, f, :, 0].sum()) == pytest.approx(float(targets.sum()), rel=1e-4)
            assert float(out[0, f, :, 1].sum()) == N
        assert float(ns2[0]) == pytest.approx(float((targets.double() ** 2).sum()), rel=1e-5)
        ref, ref_ns2 = naive_hist(codes.numpy(), nodeid.numpy(), lut.numpy(), labels.numpy(),
                                  None, 1, B, 0, None, targets=targets.numpy())
        assert np.allclose(out.numpy(), ref, atol=1e-5)
        assert np.allclose(ns2.numpy(), ref_ns2, atol=1e-4)


class TestNodeAdvance:
    def test_advance_global(self):
        # arena: tree0 root=node0 splits f0 at bin<=1 -> 1(leaf), 2 splits f1<=0 -> 3,4
        #        tree1 root=node5 is a leaf
        codes = torch.tensor([[0, 0], [2, 0], [2, 1]], dtype=torch.uint8)
        nodeid = torch.tensor([[0, 0, 0], [5, 5, 5]], dtype=torch.int32)
        feat = torch.tensor([0, -1, 1, -1, -1, -1], dtype=torch.int32)
        thr = torch.tensor([1, 0, 0, 0, 0, 0], dtype=torch.int32)
        left = torch.tensor([1, -1, 3, -1, -1, -1], dtype=torch.int32)
        right = torch.tensor([2, -1, 4, -1, -1, -1], dtype=torch.int32)
        n1 = ops.node_advance(codes, nodeid, feat, thr, left, right)
        assert n1[0].tolist() == [1, 2, 2]
        assert n1[1].tolist() == [5, 5, 5]  # keep-at-leaf: rows stay addressable
        n2 = ops.node_advance(codes, n1, feat, thr, left, right)
        assert n2[0].tolist() == [1, 3, 4]
        n3 = ops.node_advance(codes, n2, feat, thr, left, right)
        assert n3[0].tolist() == [1, 3, 4]  # stable at leaves


class TestTreePredict:
    def test_against_naive(self):
        r = np.random.default_rng(6)
        N, F, T, M, K = 30, 4, 3, 7, 2
        X = torch.from_numpy(r.normal(size=(N, F)).astype(np.float32))
        # build random valid trees: node 0 internal -> children 1,2 leaves (+ pad)
        feature = torch.full((T, M), -1, dtype=torch.int32)
        threshold = torch.zeros((T, M), dtype=torch.float32)
        left = torch.zeros((T, M), dtype=torch.int32)
        right = torch.zeros((T, M), dtype=torch.int32)
        value = torch.from_numpy(r.normal(size=(T, M, K)).astype(np.float32))
        for t in range(T):
            feature[t, 0] = int(r.integers(0, F))
            threshold[t, 0] = float(r.normal())
            left[t, 0], right[t, 0] = 1, 2
        out = ops.tree_predict(X, feature, threshold, left, right, value)
        for i in range(N):
            acc = np.zeros(K, dtype=np.float32)
            for t in range(T):
                f = int(feature[t, 0])
                node = 1 if float(X[i, f]) <= float(threshold[t, 0]) else 2
                acc += value[t, node].numpy()
            assert np.allclose(out[i].numpy(), acc, atol=1e-5)


class TestMlpGrad:
    def test_matches_autograd(self):
        torch.manual_seed(2)
        N, F, H, K = 40, 6, 9, 3
        X = torch.randn(N, F)
        y = torch.randint(0, K, (N,))
        W1 = torch.randn(H, F, requires_grad=True)
        b1 = torch.randn(H, requires_grad=True)
        W2 = torch.randn(K, H, requires_grad=True)
        b2 = torch.randn(K, requires_grad=True)
        loss, gWs, gBs = ops.mlp_grad(X, y, [W1.detach(), W2.detach()], [b1.detach(), b2.detach()])
        h = torch.relu(X @ W1.T + b1)
        ref = torch.nn.functional.cross_entropy(h @ W2.T + b2, y, reduction="sum")
        ref.backward()
        assert float(loss) == pytest.approx(float(ref.detach()), rel=1e-5)
        assert torch.allclose(gWs[0], W1.grad, atol=1e-4)
        assert torch.allclose(gBs[0], b1.grad, atol=1e-4)
        assert torch.allclose(gWs[1], W2.grad, atol=1e-4)
        assert torch.allclose(gBs[1], b2.grad, atol=1e-4)

    def test_sigmoid_activation(self):
        torch.manual_seed(3)
        N, F, H, K = 20, 4, 5, 2
        X = torch.randn(N, F)
        y = torch.randint(0, K, (N,))
        W1 = torch.randn(H, F, requires_grad=True)
        b1 = torch.zeros(H, requires_grad=True)
        W2 = torch.randn(K, H, requires_grad=True)
        b2 = torch.zeros(K, requires_grad=True)
        loss, gWs, gBs = ops.mlp_grad(
            X, y, [W1.detach(), W2.detach()], [b1.detach(), b2.detach()], activation="sigmoid"
        )
        h = torch.sigmoid(X @ W1.T + b1)
        ref = torch.nn.functional.cross_entropy(h @ W2.T + b2, y, reduction="sum")
        ref.backward()
        assert float(loss) == pytest.approx(float(ref.detach()), rel=1e-4)
        assert torch.allclose(gWs[0], W1.grad, atol=1e-4)


def test_pack_tree_nodes_encoding():
    """Packed K8 node layout: CAT_FLAG on categorical internals, float bits
    preserved, leaves stay -1."""
    from flowml import ops

    feature = torch.tensor([[2, -1, 5]], dtype=torch.int32)
    threshold = torch.tensor([[1.5, 0.0, -3.25]], dtype=torch.float32)
    left = torch.tensor([[1, -1, 0]], dtype=torch.int32)
    right = torch.tensor([[2, -1, 1]], dtype=torch.int32)
    catmask = torch.tensor([[0, 0, 0b101]], dtype=torch.int64)
    p = ops.pack_tree_nodes(feature, threshold, left, right, catmask)
    assert p.shape == (1, 3, 4) and p.dtype == torch.int32
    assert int(p[0, 0, 0]) == 2                       # continuous: bare id
    assert int(p[0, 1, 0]) == -1                      # leaf untouched
    assert int(p[0, 2, 0]) == 5 | ops.TREE_CAT_FLAG   # categorical flagged
    assert p[0, :, 1].view(torch.float32).tolist() == [1.5, 0.0, -3.25]
    assert p[0, :, 2].tolist() == [1, -1, 0]
    assert p[0, :, 3].tolist() == [2, -1, 1]
