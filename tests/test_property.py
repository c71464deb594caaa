"""Property-based tests (hypothesis) for the invariants the kernels and the
persistence layer rely on (SURVEY.md §4 "property checks")."""

import numpy as np
import pytest
import torch
from hypothesis import given, settings
from hypothesis import strategies as st

from flowml import ops
from flowml.frame import FlowFrame
from flowml.linalg import DenseVector, SparseVector, Vectors


@settings(max_examples=40, deadline=None)
@given(
    n=st.integers(1, 200),
    f=st.integers(1, 8),
    b=st.integers(2, 32),
    seed=st.integers(0, 2**31 - 1),
)
def test_bin_codes_strict_semantics(n, f, b, seed):
    """code = #(thr < x), so the binned rule `code <= k` must equal the raw
    rule `x <= thr[k]` for every bin — the train/inference consistency
    invariant the whole tree stack depends on."""
    r = np.random.default_rng(seed)
    X = torch.from_numpy(r.normal(size=(n, f)).astype(np.float32))
    # thresholds: sorted per feature with +inf padding and possible dupes
    thr = np.sort(r.normal(size=(f, b - 1)).astype(np.float32), axis=1)
    thr[r.random((f, b - 1)) < 0.2] = np.inf
    thr = np.sort(thr, axis=1)
    codes = ops.bin_codes(X, torch.from_numpy(thr)).numpy()
    Xn = X.numpy()
    for j in range(f):
        expect = (thr[j][None, :] < Xn[:, [j]]).sum(axis=1)
        assert (codes[:, j] == expect).all()
        for k in range(b - 1):
            if np.isfinite(thr[j, k]):
                assert ((codes[:, j] <= k) == (Xn[:, j] <= thr[j, k])).all()


@settings(max_examples=40, deadline=None)
@given(
    size=st.integers(1, 30),
    seed=st.integers(0, 2**31 - 1),
)
def test_sparse_dense_vector_roundtrip(size, seed):
    r = np.random.default_rng(seed)
    dense = r.normal(size=size) * (r.random(size) < 0.5)
    dv = DenseVector(dense)
    sv = Vectors.sparse(
        size,
        [int(i) for i in np.nonzero(dense)[0]],
        [float(v) for v in dense[dense != 0]],
    )
    assert np.allclose(sv.toArray(), dv.toArray())
    assert sv.numNonzeros() == int((dense != 0).sum())
    assert abs(dv.dot(dv) - float(dense @ dense)) < 1e-9
    # Spark repr round-trip shape
    sv2 = SparseVector(size, sv.indices, sv.values)
    assert np.allclose(sv2.toArray(), dense)


@settings(max_examples=30, deadline=None)
@given(
    n=st.integers(2, 500),
    seed=st.integers(0, 2**31 - 1),
    w1=st.floats(0.05, 0.95),
)
def test_random_split_partitions(n, seed, w1):
    df = FlowFrame({"x": torch.arange(n, dtype=torch.float32)})
    a, b = df.randomSplit([w1, 1 - w1], seed=seed)
    xa = a.col("x").tolist()
    xb = b.col("x").tolist()
    assert len(xa) + len(xb) == n
    assert sorted(xa + xb) == df.col("x").tolist()
    a2, b2 = df.randomSplit([w1, 1 - w1], seed=seed)
    assert a2.col("x").tolist() == xa  # seeded determinism


@settings(max_examples=30, deadline=None)
@given(
    n=st.integers(1, 300),
    k=st.integers(2, 6),
    seed=st.integers(0, 2**31 - 1),
)
def test_evaluator_invariants(n, k, seed):
    from flowml.evaluation import MulticlassClassificationEvaluator as E

    r = np.random.default_rng(seed)
    pred = torch.from_numpy(r.integers(0, k, n).astype(np.float64))
    lab = torch.from_numpy(r.integers(0, k, n).astype(np.float64))
    df = FlowFrame({"prediction": pred, "label": lab})
    acc = E(metricName="accuracy").evaluate(df)
    assert abs(acc - float((pred == lab).double().mean())) < 1e-12
    assert abs(E(metricName="hammingLoss").evaluate(df) - (1 - acc)) < 1e-12
    wr = E(metricName="weightedRecall").evaluate(df)
    assert abs(wr - acc) < 1e-12  # weighted recall == accuracy (single-label)
    f1 = E(metricName="f1").evaluate(df)
    assert 0.0 <= f1 <= 1.0


@settings(max_examples=25, deadline=None)
@given(
    n=st.integers(10, 400),
    seed=st.integers(0, 2**31 - 1),
    rate=st.floats(0.2, 1.5),
)
def test_hash_bag_cpu_reference_properties(n, seed, rate):
    """The splitmix64 bag is keyed by the GLOBAL row id: shifting the rows
    into two shards must reproduce the exact same per-row weights."""
    cdf = ops.poisson_cdf_table(rate, "poisson")
    full = ops._hash_bag_torch(2, cdf, seed, np.arange(n), 3)
    half = n // 2
    lo = ops._hash_bag_torch(2, cdf, seed, np.arange(0, half), 3)
    hi = ops._hash_bag_torch(2, cdf, seed, np.arange(half, n), 3)
    assert (np.concatenate([lo, hi]) == full).all()
    assert (full >= 0).all() and (full <= 8).all()
