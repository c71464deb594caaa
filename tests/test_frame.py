"""FlowFrame + CSV ingestion + randomSplit."""

import os

import numpy as np
import pytest
import torch

from flowml import FlowFrame, read_csv


def test_basic_columns():
    df = FlowFrame({"a": [1.0, 2.0, 3.0], "s": ["x", "y", "x"]})
    assert df.count() == 3
    assert df.columns == ["a", "s"]
    assert df.is_string("s") and not df.is_string("a")
    assert isinstance(df["a"], torch.Tensor)


def test_mismatched_lengths():
    with pytest.raises(ValueError):
        FlowFrame({"a": [1.0], "b": [1.0, 2.0]})


def test_with_column_select_drop_rename():
    df = FlowFrame({"a": [1.0, 2.0], "b": [3.0, 4.0]})
    df2 = df.withColumn("c", torch.tensor([5.0, 6.0]))
    assert df2.columns == ["a", "b", "c"]
    assert df.columns == ["a", "b"]  # immutable
    assert df2.select("c", "a").columns == ["c", "a"]
    assert df2.drop("b").columns == ["a", "c"]
    assert df2.withColumnRenamed("a", "z").columns == ["z", "b", "c"]


def test_filter_and_vector_columns():
    df = FlowFrame({"v": torch.arange(12, dtype=torch.float32).reshape(4, 3), "s": ["a", "b", "c", "d"]})
    assert df.is_vector("v")
    out = df.filter(np.array([True, False, True, False]))
    assert out.count() == 2
    assert out["s"].tolist() == ["a", "c"]
    assert torch.equal(out["v"], torch.tensor([[0.0, 1, 2], [6, 7, 8]]))


def test_random_split_seeded():
    df = FlowFrame({"a": np.arange(1000, dtype=np.float64)})
    a1, b1 = df.randomSplit([0.7, 0.3], seed=42)
    a2, b2 = df.randomSplit([0.7, 0.3], seed=42)
    assert a1.count() + b1.count() == 1000
    assert torch.equal(a1["a"], a2["a"])  # deterministic
    assert 600 < a1.count() < 800  # Binomial around 700
    a3, _ = df.randomSplit([0.7, 0.3], seed=43)
    assert not torch.equal(a1["a"], a3["a"])


def test_union_and_limit():
    df = FlowFrame({"a": [1.0, 2.0], "s": ["x", "y"]})
    u = df.union(df)
    assert u.count() == 4
    assert u.limit(3).count() == 3


def test_collect_vectors():
    df = FlowFrame({"v": torch.tensor([[1.0, 2.0]]), "a": [3.0]})
    row = df.collect()[0]
    from flowml.linalg import DenseVector

    assert isinstance(row["v"], DenseVector)
    assert row["v"].toArray().tolist() == [1.0, 2.0]
    assert row["a"] == 3.0


def test_read_csv(tmp_path):
    p = os.path.join(tmp_path, "flows.csv")
    with open(p, "w") as f:
        f.write("proto,dur,label\n")
        f.write("tcp,1.5,0\n")
        f.write("udp,2.0,1\n")
        f.write("tcp,0.5,0\n")
    df = read_csv(p)
    assert df.count() == 3
    assert df.is_string("proto")
    assert df["dur"].dtype == torch.float64
    assert df["label"].tolist() == [0, 1, 0]


def test_read_csv_no_header_schema(tmp_path):
    p = os.path.join(tmp_path, "nh.csv")
    with open(p, "w") as f:
        f.write("tcp,1.5,0\n")
        f.write("udp,2.0,1\n")
    df = read_csv(p, header=False)
    assert df.columns == ["_c0", "_c1", "_c2"]
    df2 = read_csv(p, header=False, columns=["proto", "dur", "label"])
    assert df2.columns == ["proto", "dur", "label"]


def test_streaming_classifier_cpu_fallback():
    """StreamingClassifier without a GPU: eager path, fixed-shape batches."""
    from flowml.classification import RandomForestClassifier
    from flowml.streaming import StreamingClassifier

    r = np.random.default_rng(3)
    X = torch.from_numpy(r.normal(size=(2000, 6)).astype(np.float32))
    y = (X[:, 0] > 0).to(torch.float64)
    m = RandomForestClassifier(numTrees=4, maxDepth=3, seed=0).fit(
        FlowFrame({"features": X, "label": y})
    )
    sc = StreamingClassifier(m, batch_size=500, num_features=6, device="cpu", use_graph=False)
    pred, raw = sc.infer(X[:500])
    eager = m.predictRaw(X[:500])
    assert torch.allclose(raw, eager)
    assert torch.equal(pred, eager.argmax(dim=1))


def test_parquet_round_trip(tmp_path):
    from flowml.frame import read_parquet, write_parquet

    df = FlowFrame(
        {
            "s": ["a", "b", "c"],
            "x": [1.5, 2.5, 3.5],
            "v": torch.tensor([[1.0, 2.0], [3.0, 4.0], [5.0, 6.0]]),
        }
    )
    p = str(tmp_path / "t.parquet")
    write_parquet(df, p)
    back = read_parquet(p)
    assert back["s"].tolist() == ["a", "b", "c"]
    assert torch.allclose(back["x"], df["x"])
    assert torch.allclose(back["v"], df["v"])
    sel = read_parquet(p, columns=["x"])
    assert sel.columns == ["x"]


class TestCsvMissingValues:
    def test_na_to_nan_and_imputer_pipeline(self, tmp_path):
        """Empty CSV fields load as NaN (numeric) / '' (string) and flow
        through Imputer + StringIndexer(handleInvalid) like in Spark."""
        import math

        import flowml
        from flowml.feature import Imputer, StringIndexer

        p = tmp_path / "f.csv"
        p.write_text("a,b,proto\n1.0,,tcp\n,2.5,udp\n3.0,4.5,tcp\n")
        df = flowml.read_csv(str(p))
        assert math.isnan(df.col("a")[1].item())
        assert math.isnan(df.col("b")[0].item())
        out = Imputer(inputCols=["a", "b"], outputCols=["a2", "b2"]).fit(df).transform(df)
        assert out.col("a2").tolist() == [1.0, 2.0, 3.0]
        assert out.col("b2").tolist() == [3.5, 2.5, 4.5]
        m = StringIndexer(inputCol="proto", outputCol="p").fit(out)
        assert m.transform(out).col("p").tolist() == [0.0, 1.0, 0.0]


class TestColumnMetadata:
    """Per-column metadata (the ML-attribute stand-in) must survive every
    frame transformation that keeps the column."""

    def _f(self):
        import torch
        from flowml.frame import FlowFrame

        f = FlowFrame({"a": torch.arange(6.0), "b": torch.ones(6)})
        return f.withMeta("a", {"categorical": {0: 3}})

    def test_basic_get_set(self):
        f = self._f()
        assert f.meta("a") == {"categorical": {0: 3}}
        assert f.meta("b") == {}
        # withMeta with empty dict clears
        assert f.withMeta("a", {}).meta("a") == {}
        import pytest as _pt
        with _pt.raises(KeyError):
            f.withMeta("nope", {"x": 1})

    def test_propagation(self):
        import numpy as np
        import torch

        f = self._f()
        m = {"categorical": {0: 3}}
        assert f.select("a").meta("a") == m
        assert f.drop("b").meta("a") == m
        assert f.withColumnRenamed("a", "z").meta("z") == m
        assert f.filter(torch.tensor([1, 0, 1, 0, 1, 0], dtype=torch.bool)).meta("a") == m
        assert f.take_rows(np.array([0, 2])).meta("a") == m
        assert f.limit(3).meta("a") == m
        assert f.union(f).meta("a") == m
        assert f.to("cpu").meta("a") == m
        sp = f.randomSplit([0.5, 0.5], seed=1)
        assert sp[0].meta("a") == m and sp[1].meta("a") == m

    def test_replacement_clears_stale_meta(self):
        import torch

        f = self._f()
        # replacing the column's data drops its (now stale) metadata
        assert f.withColumn("a", torch.zeros(6)).meta("a") == {}
        # other columns keep theirs
        assert f.withColumn("c", torch.zeros(6)).meta("a") != {}
        # explicit meta on withColumn sticks
        f2 = f.withColumn("c", torch.zeros(6), meta={"nominal": 4})
        assert f2.meta("c") == {"nominal": 4}

    def test_meta_is_copied_not_aliased(self):
        f = self._f()
        d = f.meta("a")
        d["categorical"][0] = 99   # nested mutation
        d["new"] = 1               # top-level mutation
        assert f.meta("a") == {"categorical": {0: 3}}  # deep-copied both ways
