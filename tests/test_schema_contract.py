"""Arrow-type-level save-format contract (SURVEY.md §2.2 persistence row).

The round-trip tests prove flowml can read its own output; these assert the
PHYSICAL parquet/JSON types with raw pyarrow, independent of the loader —
drift in the writer fails here even when save→load still round-trips.
"""

import json
import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from flowml import FlowFrame
from flowml.classification import LogisticRegression, RandomForestClassifier


def _df(seed=0, n=300, k=3):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, 5)).astype(np.float32)
    y = r.integers(0, k, n).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y})


def _read_one(path):
    files = [f for f in os.listdir(path) if f.endswith(".parquet")]
    assert files, path
    return pq.read_table(os.path.join(path, files[0]))


class TestArrowTypes:
    def test_vector_and_matrix_udt_physical_types(self, tmp_path):
        m = LogisticRegression(maxIter=5).fit(_df())
        p = str(tmp_path / "lr")
        m.save(p)
        t = _read_one(os.path.join(p, "data"))
        vec = t.schema.field("interceptVector").type
        assert pa.types.is_struct(vec)
        fields = {vec.field(i).name: vec.field(i).type for i in range(vec.num_fields)}
        # Spark VectorUDT: type tinyint, size int, indices array<int>,
        # values array<double>
        assert set(fields) == {"type", "size", "indices", "values"}
        assert pa.types.is_integer(fields["type"])
        assert pa.types.is_list(fields["values"])
        assert pa.types.is_float64(fields["values"].value_type)
        mat = t.schema.field("coefficientMatrix").type
        mfields = {mat.field(i).name: mat.field(i).type for i in range(mat.num_fields)}
        assert set(mfields) == {"type", "numRows", "numCols", "colPtrs",
                                "rowIndices", "values", "isTransposed"}
        assert pa.types.is_boolean(mfields["isTransposed"])
        assert pa.types.is_float64(mfields["values"].value_type)
        # scalars
        assert pa.types.is_integer(t.schema.field("numClasses").type)
        assert pa.types.is_boolean(t.schema.field("isMultinomial").type)

    def test_node_data_physical_types(self, tmp_path):
        m = RandomForestClassifier(numTrees=2, maxDepth=2, seed=0).fit(_df())
        p = str(tmp_path / "rf")
        m.save(p)
        t = _read_one(os.path.join(p, "data"))
        assert pa.types.is_integer(t.schema.field("treeID").type)
        nd = t.schema.field("nodeData").type
        f = {nd.field(i).name: nd.field(i).type for i in range(nd.num_fields)}
        assert pa.types.is_float64(f["prediction"])
        assert pa.types.is_float64(f["impurity"])
        assert pa.types.is_list(f["impurityStats"])
        assert pa.types.is_float64(f["impurityStats"].value_type)
        assert pa.types.is_integer(f["leftChild"])
        sp = f["split"]
        sf = {sp.field(i).name: sp.field(i).type for i in range(sp.num_fields)}
        assert pa.types.is_list(sf["leftCategoriesOrThreshold"])
        assert pa.types.is_float64(sf["leftCategoriesOrThreshold"].value_type)
        assert pa.types.is_integer(sf["numCategories"])
        # treesMetadata: treeID int, metadata string(json), weights double
        tm = _read_one(os.path.join(p, "treesMetadata"))
        assert pa.types.is_string(tm.schema.field("metadata").type)
        assert pa.types.is_float64(tm.schema.field("weights").type)
        json.loads(tm.to_pylist()[0]["metadata"])  # valid JSON

    def test_metadata_is_single_json_line(self, tmp_path):
        m = LogisticRegression(maxIter=5).fit(_df())
        p = str(tmp_path / "lrm")
        m.save(p)
        raw = open(os.path.join(p, "metadata", "part-00000")).read()
        assert raw.endswith("\n") and raw.count("\n") == 1  # one-line JSON
        meta = json.loads(raw)
        assert set(meta) == {"class", "timestamp", "sparkVersion", "uid",
                             "paramMap", "defaultParamMap"}
        assert isinstance(meta["timestamp"], int)  # epoch millis
        assert meta["timestamp"] > 1_500_000_000_000
        # param values are plain JSON scalars/lists (no numpy leakage)
        for v in list(meta["paramMap"].values()) + list(meta["defaultParamMap"].values()):
            assert isinstance(v, (int, float, str, bool, list, type(None)))

    def test_success_markers_everywhere(self, tmp_path):
        m = RandomForestClassifier(numTrees=2, maxDepth=2, seed=0).fit(_df())
        p = str(tmp_path / "rfs")
        m.save(p)
        for sub in ("metadata", "data", "treesMetadata"):
            assert os.path.isfile(os.path.join(p, sub, "_SUCCESS")), sub
