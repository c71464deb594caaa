"""FlowFrame — eager columnar flow-record batches over torch tensors.

This is the MI355X-native replacement for the Spark SQL DataFrame layer the
reference drives: no Catalyst plan, no lazy evaluation —
a FlowFrame is a dict of resident columns. Numeric columns are torch tensors
(GPU-resident when a device is set, sized for 288 GB HBM3E per GPU); string
(categorical) columns are host numpy arrays (GPUs don't hold strings — the
StringIndexer maps them to double index columns at featurization, after which
everything on the hot path is a tensor). Vector columns (e.g. the
VectorAssembler output) are 2-D tensors of shape (n_rows, dim).

CSV ingestion (``read_csv``) uses pyarrow's multithreaded C++ CSV reader,
replacing ``spark.read.csv`` (SURVEY.md §2.2 "CSV ingestion").
"""

from __future__ import annotations

from typing import Any, Dict, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

Column = Union[torch.Tensor, np.ndarray]


class FlowFrame:
    """An eager columnar batch: ordered ``{name: column}`` with equal lengths."""

    def __init__(self, columns: "Dict[str, Any]", device: Optional[Union[str, torch.device]] = None,
                 meta: Optional[Dict[str, dict]] = None):
        self._cols: Dict[str, Column] = {}
        self._nrows: Optional[int] = None
        # per-column metadata (the stand-in for Spark's ML column attributes:
        # e.g. {"categorical": {featIdx: arity}} on a features vector, or
        # {"nominal": numLabels} on a StringIndexer output column)
        import copy as _copy

        self._meta: Dict[str, dict] = {
            k: _copy.deepcopy(v) for k, v in (meta or {}).items()
            if k in columns and v
        }
        self.device = torch.device(device) if device is not None else torch.device("cpu")
        # device is honored, not just recorded: tensor columns are moved so
        # FlowFrame({...}, device="cuda") behaves like .to("cuda") (string
        # columns stay host-side numpy by design)
        move = device is not None
        for name, col in columns.items():
            c = self._coerce(col)
            if move and isinstance(c, torch.Tensor) and c.device != self.device:
                c = c.to(self.device, non_blocking=True)
            self._cols[name] = c
        for name, col in self._cols.items():
            n = len(col)
            if self._nrows is None:
                self._nrows = n
            elif self._nrows != n:
                raise ValueError(
                    f"column {name!r} has {n} rows, expected {self._nrows}"
                )
        if self._nrows is None:
            self._nrows = 0

    # -- construction helpers ----------------------------------------------

    def _coerce(self, col: Any) -> Column:
        if isinstance(col, torch.Tensor):
            return col
        arr = np.asarray(col)
        if arr.dtype.kind in "UOS":  # strings stay numpy on host
            return arr.astype(object)
        if arr.dtype.kind == "b":
            return torch.from_numpy(arr.astype(np.bool_))
        return torch.from_numpy(np.ascontiguousarray(arr))

    @staticmethod
    def from_dict(data: Dict[str, Any], device=None) -> "FlowFrame":
        return FlowFrame(data, device=device)

    # -- column metadata ------------------------------------------------------

    def meta(self, name: str) -> dict:
        """Metadata dict attached to a column ({} if none). Deep-copied —
        mutating the result never changes the frame."""
        import copy as _copy

        return _copy.deepcopy(self._meta.get(name, {}))

    def withMeta(self, name: str, meta: dict) -> "FlowFrame":
        """Same frame with ``meta`` attached to column ``name``."""
        if name not in self._cols:
            raise KeyError(name)
        m = dict(self._meta)
        if meta:
            m[name] = dict(meta)
        else:
            m.pop(name, None)
        return FlowFrame(self._cols, device=self.device, meta=m)

    # -- basic introspection -------------------------------------------------

    @property
    def columns(self) -> List[str]:
        return list(self._cols.keys())

    def count(self) -> int:
        return self._nrows or 0

    def __len__(self) -> int:
        return self.count()

    def __contains__(self, name: str) -> bool:
        return name in self._cols

    def __getitem__(self, name: str) -> Column:
        return self._cols[name]

    def col(self, name: str) -> Column:
        if name not in self._cols:
            raise KeyError(f"Column {name!r} not found; available: {self.columns}")
        return self._cols[name]

    def is_string(self, name: str) -> bool:
        return isinstance(self._cols[name], np.ndarray)

    def is_vector(self, name: str) -> bool:
        c = self._cols[name]
        return isinstance(c, torch.Tensor) and c.dim() == 2

    def dtypes(self) -> List[Tuple[str, str]]:
        out = []
        for name, col in self._cols.items():
            if isinstance(col, np.ndarray):
                out.append((name, "string"))
            elif col.dim() == 2:
                out.append((name, "vector"))
            else:
                out.append((name, str(col.dtype).replace("torch.", "")))
        return out

    # -- transformations -----------------------------------------------------

    def withColumn(self, name: str, col: Any, meta: Optional[dict] = None) -> "FlowFrame":
        cols = dict(self._cols)
        cols[name] = col
        m = {k: v for k, v in self._meta.items() if k != name}
        if meta:
            m[name] = dict(meta)
        return FlowFrame(cols, device=self.device, meta=m)

    def select(self, *names: str) -> "FlowFrame":
        flat: List[str] = []
        for n in names:
            if isinstance(n, (list, tuple)):
                flat.extend(n)
            else:
                flat.append(n)
        return FlowFrame({n: self.col(n) for n in flat}, device=self.device, meta=self._meta)

    def drop(self, *names: str) -> "FlowFrame":
        return FlowFrame(
            {n: c for n, c in self._cols.items() if n not in names},
            device=self.device, meta=self._meta,
        )

    def withColumnRenamed(self, old: str, new: str) -> "FlowFrame":
        cols = {}
        for n, c in self._cols.items():
            cols[new if n == old else n] = c
        m = {(new if k == old else k): v for k, v in self._meta.items()}
        return FlowFrame(cols, device=self.device, meta=m)

    def filter(self, mask: Union[torch.Tensor, np.ndarray]) -> "FlowFrame":
        """Row subset by boolean mask (tensor on any device, or numpy)."""
        if isinstance(mask, torch.Tensor):
            mask_np = mask.cpu().numpy().astype(bool)
        else:
            mask_np = np.asarray(mask, dtype=bool)
        return self._take_mask(mask_np)

    def _take_mask(self, mask_np: np.ndarray) -> "FlowFrame":
        idx_t: Optional[torch.Tensor] = None
        cols: Dict[str, Column] = {}
        for n, c in self._cols.items():
            if isinstance(c, np.ndarray):
                cols[n] = c[mask_np]
            else:
                if idx_t is None or idx_t.device != c.device:
                    idx_t = torch.from_numpy(np.nonzero(mask_np)[0]).to(c.device)
                cols[n] = c.index_select(0, idx_t)
        return FlowFrame(cols, device=self.device, meta=self._meta)

    def take_rows(self, indices: np.ndarray) -> "FlowFrame":
        cols: Dict[str, Column] = {}
        idx_cache: Dict[torch.device, torch.Tensor] = {}
        for n, c in self._cols.items():
            if isinstance(c, np.ndarray):
                cols[n] = c[indices]
            else:
                t = idx_cache.get(c.device)
                if t is None:
                    t = torch.from_numpy(np.ascontiguousarray(indices)).to(c.device)
                    idx_cache[c.device] = t
                cols[n] = c.index_select(0, t)
        return FlowFrame(cols, device=self.device, meta=self._meta)

    def limit(self, n: int) -> "FlowFrame":
        return self.take_rows(np.arange(min(n, self.count()), dtype=np.int64))

    def union(self, other: "FlowFrame") -> "FlowFrame":
        if self.columns != other.columns:
            raise ValueError("union requires identical column sets/order")
        cols: Dict[str, Column] = {}
        for n in self.columns:
            a, b = self._cols[n], other._cols[n]
            if isinstance(a, np.ndarray):
                cols[n] = np.concatenate([a, np.asarray(b, dtype=object)])
            else:
                cols[n] = torch.cat([a, b], dim=0)
        return FlowFrame(cols, device=self.device, meta=self._meta)

    # -- randomSplit ---------------------------------------------------------

    def randomSplit(self, weights: Sequence[float], seed: Optional[int] = None) -> List["FlowFrame"]:
        """Seeded per-row split, semantics of ``DataFrame.randomSplit``.

        Spark assigns each row an independent uniform draw and buckets it by
        normalized cumulative weights (SURVEY.md §2.2 "train/test utilities") —
        split sizes are therefore Binomial, not exact fractions. Reproduced
        with a numpy PCG64 stream keyed by ``seed``.
        """
        w = np.asarray(weights, dtype=np.float64)
        if np.any(w < 0):
            raise ValueError("weights must be nonnegative")
        cum = np.cumsum(w / w.sum())
        rng = np.random.default_rng(None if seed is None else seed)
        u = rng.random(self.count())
        out = []
        lo = 0.0
        for hi in cum:
            mask = (u >= lo) & (u < hi) if hi < 1.0 else (u >= lo)
            out.append(self._take_mask(mask))
            lo = hi
        return out

    # -- device movement -----------------------------------------------------

    def to(self, device: Union[str, torch.device]) -> "FlowFrame":
        device = torch.device(device)
        cols: Dict[str, Column] = {}
        for n, c in self._cols.items():
            cols[n] = c if isinstance(c, np.ndarray) else c.to(device, non_blocking=True)
        return FlowFrame(cols, device=device, meta=self._meta)

    def cpu(self) -> "FlowFrame":
        return self.to("cpu")

    # -- materialization -----------------------------------------------------

    def toPandas(self):
        import pandas as pd

        data = {}
        for n, c in self._cols.items():
            if isinstance(c, np.ndarray):
                data[n] = c
            elif c.dim() == 2:
                data[n] = list(c.detach().cpu().numpy())
            else:
                data[n] = c.detach().cpu().numpy()
        return pd.DataFrame(data)

    def collect(self) -> List[Dict[str, Any]]:
        """Rows as python dicts (vector columns become DenseVector)."""
        from flowml.linalg import DenseVector

        mats = {}
        for n, c in self._cols.items():
            if isinstance(c, np.ndarray):
                mats[n] = c
            else:
                mats[n] = c.detach().cpu().numpy()
        rows = []
        for i in range(self.count()):
            row = {}
            for n, m in mats.items():
                v = m[i]
                if isinstance(v, np.ndarray) and v.ndim == 1:
                    row[n] = DenseVector(v)
                elif isinstance(v, (np.floating,)):
                    row[n] = float(v)
                elif isinstance(v, (np.integer,)):
                    row[n] = int(v)
                elif isinstance(v, (np.bool_,)):
                    row[n] = bool(v)
                else:
                    row[n] = v
            rows.append(row)
        return rows

    def head(self, n: int = 5) -> List[Dict[str, Any]]:
        return self.limit(n).collect()

    def show(self, n: int = 20) -> None:
        for row in self.head(n):
            print(row)

    def __repr__(self) -> str:
        kinds = ", ".join(f"{n}: {t}" for n, t in self.dtypes())
        return f"FlowFrame[{self.count()} rows]({kinds})"


# ---------------------------------------------------------------------------
# CSV I/O
# ---------------------------------------------------------------------------


def read_csv(
    path: str,
    header: bool = True,
    inferSchema: bool = True,
    schema: Optional[Dict[str, str]] = None,
    columns: Optional[List[str]] = None,
    device=None,
) -> FlowFrame:
    """Typed columnar CSV load (``spark.read.csv`` equivalent).

    Uses pyarrow's parallel CSV reader. ``schema`` maps column name to one of
    ``{"double","float","int","long","string"}``; without it and with
    ``inferSchema=True`` pyarrow infers types. ``columns`` names columns when
    the file has no header row (Spark's ``_c0.._cN`` default naming is applied
    otherwise).
    """
    import pyarrow as pa
    import pyarrow.csv as pacsv

    read_opts = pacsv.ReadOptions(autogenerate_column_names=not header)
    if columns is not None and not header:
        read_opts = pacsv.ReadOptions(column_names=list(columns))
    convert_opts = pacsv.ConvertOptions()
    if schema is not None:
        type_map = {
            "double": pa.float64(),
            "float": pa.float32(),
            "int": pa.int32(),
            "long": pa.int64(),
            "string": pa.string(),
        }
        convert_opts = pacsv.ConvertOptions(
            column_types={k: type_map[v] for k, v in schema.items()}
        )
    table = pacsv.read_csv(path, read_options=read_opts, convert_options=convert_opts)
    if not header and columns is None:
        table = table.rename_columns([f"_c{i}" for i in range(table.num_columns)])
    return from_arrow(table, device=device)


def read_parquet(path: str, columns: Optional[List[str]] = None, device=None) -> FlowFrame:
    """Columnar Parquet load (pyarrow C++ reader; SURVEY.md L3 'CSV/Parquet I/O')."""
    import pyarrow.parquet as pq

    table = pq.read_table(path, columns=columns)
    return from_arrow(table, device=device)


def write_parquet(df: FlowFrame, path: str, compression: str = "zstd") -> None:
    """Write a FlowFrame to Parquet (vector columns as FixedSizeList)."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    arrays = []
    names = []
    for n in df.columns:
        c = df[n]
        if isinstance(c, np.ndarray):
            arrays.append(pa.array(c.tolist(), type=pa.string()))
        else:
            t = c.detach().cpu()
            if t.dim() == 2:
                flat = pa.array(t.reshape(-1).numpy())
                arrays.append(pa.FixedSizeListArray.from_arrays(flat, t.shape[1]))
            else:
                arrays.append(pa.array(t.numpy()))
        names.append(n)
    pq.write_table(pa.table(arrays, names=names), path, compression=compression)


def from_arrow(table, device=None) -> FlowFrame:
    """Build a FlowFrame from a pyarrow Table (zero-copy where possible)."""
    import pyarrow as pa

    cols: Dict[str, Any] = {}
    for name, col in zip(table.column_names, table.columns):
        combined = col.combine_chunks()
        if pa.types.is_string(combined.type) or pa.types.is_large_string(combined.type):
            cols[name] = np.asarray(combined.to_pylist(), dtype=object)
        elif pa.types.is_fixed_size_list(combined.type) or pa.types.is_list(combined.type):
            np_arr = np.stack(combined.to_numpy(zero_copy_only=False))
            cols[name] = torch.from_numpy(np.ascontiguousarray(np_arr))
        else:
            np_arr = combined.to_numpy(zero_copy_only=False)
            # pyarrow hands back a read-only view; copy so torch owns
            # writable memory (silences the non-writable-tensor warning)
            cols[name] = torch.from_numpy(
                np.ascontiguousarray(np_arr).copy()
                if not np_arr.flags.writeable else np.ascontiguousarray(np_arr)
            )
    ff = FlowFrame(cols)
    if device is not None:
        ff = ff.to(device)
    return ff
