"""Featurization stages: StringIndexer, OneHotEncoder, VectorAssembler, scalers.

Behavioral spec: ``pyspark.ml.feature`` (upstream ``spark:.../ml/feature/``;
SURVEY.md §2.2 rows 5-8). Semantics that matter for parity, all replicated:

* StringIndexer orders labels by **descending frequency, ties broken
  alphabetically** by default (``stringOrderType="frequencyDesc"``);
  ``handleInvalid`` ∈ {error (default), skip, keep} with keep mapping unseen
  labels to index ``numLabels``.
* OneHotEncoder drops the last category by default (``dropLast=True``) so the
  encoding has ``numCategories - 1`` slots; ``handleInvalid="keep"`` adds one
  extra slot for invalid/unseen.
* VectorAssembler concatenates numeric and vector columns in declared order.
* StandardScaler defaults to ``withMean=False, withStd=True`` (!) with the
  sample (n-1) standard deviation; zero-variance features scale to 0.
* MinMaxScaler maps to [min,max] with constant features at the midpoint.

MI355X note: the assembled feature column is a dense 2-D tensor (row-major,
N×D) rather than per-row sparse vectors — at NSL-KDD shape (41 features →
~120 after one-hot) dense HBM-resident columns are strictly better for the
downstream kernels. The user-visible per-row view (``collect``) still yields
``flowml.linalg`` vectors.
"""

from __future__ import annotations

import math
from typing import Dict, List, Optional

import numpy as np
import torch

from flowml.base import Estimator, Model, Transformer
from flowml.frame import FlowFrame
from flowml.param import (
    HasHandleInvalid,
    HasInputCol,
    HasInputCols,
    HasLabelCol,
    HasOutputCol,
    HasOutputCols,
    Param,
    Params,
    TypeConverters,
)


# ---------------------------------------------------------------------------
# StringIndexer
# ---------------------------------------------------------------------------


def _column_as_strings(df: FlowFrame, name: str) -> np.ndarray:
    col = df.col(name)
    if isinstance(col, np.ndarray):
        return col.astype(str)
    # Spark casts numeric input columns to string before indexing
    vals = col.detach().cpu().numpy()
    return np.asarray([str(v) for v in vals], dtype=object).astype(str)


def _order_labels(values: np.ndarray, counts: np.ndarray, order: str) -> List[str]:
    if order == "frequencyDesc":
        idx = np.lexsort((values, -counts))
    elif order == "frequencyAsc":
        idx = np.lexsort((values, counts))
    elif order == "alphabetAsc":
        idx = np.argsort(values, kind="stable")
    elif order == "alphabetDesc":
        idx = np.argsort(values, kind="stable")[::-1]
    else:
        raise ValueError(f"Unsupported stringOrderType: {order}")
    return [str(v) for v in values[idx]]


class _StringIndexerParams(HasInputCol, HasInputCols, HasOutputCol, HasOutputCols, HasHandleInvalid):
    stringOrderType: Param[str] = Param(
        "undefined",
        "stringOrderType",
        "How to order labels of string column. The first label after ordering is assigned an index of 0. "
        "Supported options: frequencyDesc, frequencyAsc, alphabetDesc, alphabetAsc.",
        TypeConverters.toString,
    )

    def __init__(self):
        super().__init__()
        self._setDefault(handleInvalid="error", stringOrderType="frequencyDesc")

    def getStringOrderType(self) -> str:
        return self.getOrDefault(self.stringOrderType)

    def _in_out_cols(self):
        if self.isSet(self.inputCols):
            return list(self.getInputCols()), list(self.getOrDefault(self.outputCols))
        return [self.getInputCol()], [self.getOutputCol()]


class StringIndexer(_StringIndexerParams, Estimator):
    """Map string labels to double indices by descending frequency.

    >>> import numpy as np
    >>> from flowml.frame import FlowFrame
    >>> df = FlowFrame({"s": np.array(["b", "a", "b", "c", "b", "a"], dtype=object)})
    >>> m = StringIndexer(inputCol="s", outputCol="i").fit(df)
    >>> m.labels
    ['b', 'a', 'c']
    >>> m.transform(df)["i"].tolist()
    [0.0, 1.0, 0.0, 2.0, 0.0, 1.0]
    """

    """Label → index by frequency (SURVEY.md §2.2; spark:.../feature/StringIndexer.scala)."""

    def __init__(
        self,
        *,
        inputCol: Optional[str] = None,
        outputCol: Optional[str] = None,
        inputCols: Optional[List[str]] = None,
        outputCols: Optional[List[str]] = None,
        handleInvalid: Optional[str] = None,
        stringOrderType: Optional[str] = None,
    ):
        super().__init__()
        self._set(
            inputCol=inputCol,
            outputCol=outputCol,
            inputCols=inputCols,
            outputCols=outputCols,
            handleInvalid=handleInvalid,
            stringOrderType=stringOrderType,
        )

    def setStringOrderType(self, value: str):
        return self.set(self.stringOrderType, value)

    def _fit(self, dataset: FlowFrame) -> "StringIndexerModel":
        from flowml import dist

        in_cols, _ = self._in_out_cols()
        order = self.getStringOrderType()
        labelsArray: List[List[str]] = []
        for c in in_cols:
            vals = _column_as_strings(dataset, c)
            uniq, counts = np.unique(vals, return_counts=True)
            if dist.get_world_size() > 1:
                # merge per-rank frequency dicts so every
                # rank derives the identical label order on sharded data
                merged: Dict[str, int] = {}
                for part in dist.all_gather_object(
                    dict(zip(uniq.tolist(), (int(v) for v in counts)))
                ):
                    for k, v in part.items():
                        merged[k] = merged.get(k, 0) + v
                uniq = np.array(sorted(merged.keys()), dtype=object)
                counts = np.array([merged[k] for k in uniq], dtype=np.int64)
            labelsArray.append(_order_labels(uniq, counts, order))
        model = StringIndexerModel(labelsArray=labelsArray)
        self._copyValues(model)
        model._set_parent(self)
        return model


class StringIndexerModel(_StringIndexerParams, Model):
    def __init__(self, labelsArray: Optional[List[List[str]]] = None):
        super().__init__()
        self.labelsArray: List[List[str]] = labelsArray or []

    @property
    def labels(self) -> List[str]:
        return self.labelsArray[0] if self.labelsArray else []

    @classmethod
    def from_labels(cls, labels: List[str], inputCol: str, outputCol: str, handleInvalid="error"):
        m = cls(labelsArray=[list(labels)])
        m._set(inputCol=inputCol, outputCol=outputCol, handleInvalid=handleInvalid)
        return m

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        in_cols, out_cols = self._in_out_cols()
        handle = self.getHandleInvalid()
        keep_mask: Optional[np.ndarray] = None
        new_cols: Dict[str, np.ndarray] = {}
        for c, o, labels in zip(in_cols, out_cols, self.labelsArray):
            vals = _column_as_strings(dataset, c)
            lookup = {lab: float(i) for i, lab in enumerate(labels)}
            idx = np.fromiter(
                (lookup.get(v, -1.0) for v in vals), dtype=np.float64, count=len(vals)
            )
            invalid = idx < 0
            if invalid.any():
                if handle == "error":
                    bad = sorted(set(np.asarray(vals)[invalid].tolist()))[:5]
                    raise ValueError(
                        f"StringIndexer encountered unseen labels {bad} in column {c!r} "
                        f"(handleInvalid='error'). Use 'skip' or 'keep'."
                    )
                elif handle == "skip":
                    keep_mask = ~invalid if keep_mask is None else (keep_mask & ~invalid)
                elif handle == "keep":
                    idx = np.where(invalid, float(len(labels)), idx)
                else:
                    raise ValueError(f"Unsupported handleInvalid: {handle}")
            new_cols[o] = idx
        out = dataset
        for o, idx, labels in zip(
            new_cols.keys(), new_cols.values(), self.labelsArray
        ):
            # nominal attribute (Spark NominalAttribute.withValues): lets
            # VectorAssembler declare this slot categorical for the trees;
            # handleInvalid='keep' adds one extra index = numLabels
            n_vals = len(labels) + (1 if handle == "keep" else 0)
            out = out.withColumn(
                o, torch.from_numpy(idx).to(dataset.device),
                meta={"nominal": n_vals, "labels": list(labels)},
            )
        if keep_mask is not None:
            out = out.filter(keep_mask)
        return out

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# OneHotEncoder
# ---------------------------------------------------------------------------


class _OneHotEncoderParams(HasInputCol, HasInputCols, HasOutputCol, HasOutputCols, HasHandleInvalid):
    dropLast: Param[bool] = Param(
        "undefined", "dropLast", "whether to drop the last category", TypeConverters.toBoolean
    )

    def __init__(self):
        super().__init__()
        self._setDefault(handleInvalid="error", dropLast=True)

    def getDropLast(self) -> bool:
        return self.getOrDefault(self.dropLast)

    def _in_out_cols(self):
        if self.isSet(self.inputCols):
            return list(self.getInputCols()), list(self.getOrDefault(self.outputCols))
        return [self.getInputCol()], [self.getOutputCol()]


class OneHotEncoder(_OneHotEncoderParams, Estimator):
    """Index → one-hot vector, dropLast=True default (spark:.../feature/OneHotEncoder.scala)."""

    def __init__(
        self,
        *,
        inputCol: Optional[str] = None,
        outputCol: Optional[str] = None,
        inputCols: Optional[List[str]] = None,
        outputCols: Optional[List[str]] = None,
        handleInvalid: Optional[str] = None,
        dropLast: Optional[bool] = None,
    ):
        super().__init__()
        self._set(
            inputCol=inputCol,
            outputCol=outputCol,
            inputCols=inputCols,
            outputCols=outputCols,
            handleInvalid=handleInvalid,
            dropLast=dropLast,
        )

    def setDropLast(self, value: bool):
        return self.set(self.dropLast, value)

    def _fit(self, dataset: FlowFrame) -> "OneHotEncoderModel":
        in_cols, _ = self._in_out_cols()
        sizes = []
        for c in in_cols:
            col = dataset.col(c)
            if isinstance(col, np.ndarray):
                raise TypeError(f"OneHotEncoder input {c!r} must be numeric (use StringIndexer first)")
            mx = int(col.max().item()) if len(col) else -1
            if len(col) and float(col.min().item()) < 0:
                raise ValueError(f"OneHotEncoder input {c!r} has negative values")
            sizes.append(mx + 1)
        from flowml import dist

        if dist.get_world_size() > 1:
            sizes = [max(s) for s in zip(*dist.all_gather_object(sizes))]
        model = OneHotEncoderModel(categorySizes=sizes)
        self._copyValues(model)
        model._set_parent(self)
        return model


class OneHotEncoderModel(_OneHotEncoderParams, Model):
    def __init__(self, categorySizes: Optional[List[int]] = None):
        super().__init__()
        self.categorySizes: List[int] = categorySizes or []

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        in_cols, out_cols = self._in_out_cols()
        handle = self.getHandleInvalid()
        drop_last = self.getDropLast()
        out = dataset
        keep_mask: Optional[torch.Tensor] = None
        for c, o, n_cats in zip(in_cols, out_cols, self.categorySizes):
            col = dataset.col(c)
            if isinstance(col, np.ndarray):
                raise TypeError(f"OneHotEncoder input {c!r} must be numeric")
            idx = col.to(torch.int64)
            # vector width: numCats (+1 if keep) (-1 if dropLast)
            size = n_cats + (1 if handle == "keep" else 0) - (1 if drop_last else 0)
            invalid = (idx < 0) | (idx >= n_cats)
            if bool(invalid.any()):
                if handle == "error":
                    raise ValueError(
                        f"OneHotEncoder found values out of range [0,{n_cats}) in {c!r} "
                        f"(handleInvalid='error')."
                    )
                elif handle == "keep":
                    idx = torch.where(invalid, torch.full_like(idx, n_cats), idx)
                elif handle == "skip":
                    m = ~invalid
                    keep_mask = m if keep_mask is None else (keep_mask & m)
                    idx = torch.where(invalid, torch.zeros_like(idx), idx)  # placeholder
                else:
                    raise ValueError(f"Unsupported handleInvalid: {handle}")
            enc = torch.zeros((len(idx), size), dtype=torch.float32, device=idx.device)
            in_range = idx < size  # dropLast: last category encodes as all-zeros
            rows = torch.nonzero(in_range, as_tuple=False).flatten()
            if rows.numel():
                enc[rows, idx[rows]] = 1.0
            out = out.withColumn(o, enc)
        if keep_mask is not None:
            out = out.filter(keep_mask)
        return out

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# VectorAssembler
# ---------------------------------------------------------------------------


class VectorAssembler(HasInputCols, HasOutputCol, HasHandleInvalid, Transformer):
    """Concatenate numeric + vector columns into one feature vector column.

    Spec: spark:.../feature/VectorAssembler.scala.
    Output is a dense 2-D float32 tensor column (see module docstring).
    handleInvalid: 'error' raises on NaN, 'skip' drops rows with NaN, 'keep'
    passes NaN through.
    """

    def __init__(
        self,
        *,
        inputCols: Optional[List[str]] = None,
        outputCol: Optional[str] = None,
        handleInvalid: Optional[str] = None,
    ):
        super().__init__()
        self._setDefault(handleInvalid="error")
        self._set(inputCols=inputCols, outputCol=outputCol, handleInvalid=handleInvalid)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        in_cols = self.getInputCols()
        out_col = self.getOutputCol()
        handle = self.getHandleInvalid()
        parts: List[torch.Tensor] = []
        device = None
        categorical: Dict[int, int] = {}
        offset = 0
        for c in in_cols:
            col = dataset.col(c)
            if isinstance(col, np.ndarray):
                raise TypeError(
                    f"VectorAssembler input {c!r} is a string column; index it first"
                )
            t = col if col.dim() == 2 else col.reshape(-1, 1)
            # merge ML attributes (Spark AttributeGroup): a nominal scalar
            # input (StringIndexer output) or a categorical vector input
            # (VectorIndexer output) marks its slots categorical
            cmeta = dataset.meta(c)
            if t.shape[1] == 1 and cmeta.get("nominal"):
                categorical[offset] = int(cmeta["nominal"])
            for j, a in (cmeta.get("categorical") or {}).items():
                categorical[offset + int(j)] = int(a)
            offset += t.shape[1]
            parts.append(t.to(torch.float32))
            device = t.device
        feats = torch.cat(parts, dim=1) if len(parts) > 1 else parts[0].clone()
        out_meta = {"categorical": categorical} if categorical else None
        if handle != "keep":
            bad = torch.isnan(feats).any(dim=1)
            if bool(bad.any()):
                if handle == "error":
                    raise ValueError(
                        "VectorAssembler encountered NaN/null values (handleInvalid='error')."
                    )
                out = dataset.withColumn(out_col, feats, meta=out_meta)
                return out.filter(~bad)
        return dataset.withColumn(out_col, feats, meta=out_meta)


# ---------------------------------------------------------------------------
# VectorIndexer (spark:.../feature/VectorIndexer.scala)
# ---------------------------------------------------------------------------


class _VectorIndexerParams(HasInputCol, HasOutputCol, HasHandleInvalid):
    maxCategories: Param[int] = Param(
        "undefined", "maxCategories",
        "Threshold for the number of values a categorical feature can take "
        "(>= 2). If a feature is found to have > maxCategories values, then "
        "it is declared continuous.",
        TypeConverters.toInt,
    )

    def __init__(self):
        super().__init__()
        self._setDefault(maxCategories=20, handleInvalid="error")

    def getMaxCategories(self) -> int:
        return self.getOrDefault(self.maxCategories)


class VectorIndexer(_VectorIndexerParams, Estimator):
    """Automatically decide which features are categorical and index them.

    >>> import torch
    >>> from flowml.frame import FlowFrame
    >>> X = torch.tensor([[10.0, 0.3], [0.0, -1.2], [10.0, 2.4], [5.0, 0.7]])
    >>> m = VectorIndexer(inputCol="f", outputCol="ix", maxCategories=3).fit(
    ...     FlowFrame({"f": X}))
    >>> m.categoryMaps
    {0: {0.0: 0, 5.0: 1, 10.0: 2}}
    >>> out = m.transform(FlowFrame({"f": X}))
    >>> out["ix"][:, 0].tolist()
    [2.0, 0.0, 2.0, 1.0]
    >>> out.meta("ix")
    {'categorical': {0: 3}}

    Spark semantics (spark:.../feature/VectorIndexer.scala): a feature with
    at most ``maxCategories`` distinct values is declared categorical; its
    values are re-indexed with **0.0 kept at index 0** (to preserve
    sparsity) and the remaining values sorted ascending. The output column
    carries categorical metadata ({"categorical": {featIdx: arity}}) that
    the tree estimators read — the stand-in for Spark's ML attributes.
    Distributed: per-rank distinct sets are merged across ranks so every
    rank fits the identical model.
    """

    def __init__(self, *, inputCol: Optional[str] = None,
                 outputCol: Optional[str] = None,
                 maxCategories: Optional[int] = None,
                 handleInvalid: Optional[str] = None):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol,
                  maxCategories=maxCategories, handleInvalid=handleInvalid)

    def setMaxCategories(self, v: int):
        return self.set(self.maxCategories, v)

    def _fit(self, dataset: FlowFrame) -> "VectorIndexerModel":
        from flowml import dist

        x = dataset.col(self.getInputCol())
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("VectorIndexer input must be a vector column")
        maxc = self.getMaxCategories()
        if maxc < 2:
            raise ValueError("maxCategories must be >= 2")
        F = x.shape[1]
        xd = x.to(torch.float64)
        distinct: List[Optional[np.ndarray]] = []
        for j in range(F):
            u = torch.unique(xd[:, j])
            # cutoff: once a feature exceeds maxCategories it is continuous;
            # keep maxc+1 values so the merged set can still exceed the cap
            distinct.append(u.cpu().numpy()[: maxc + 1] if len(u) <= maxc else None)
        if dist.get_world_size() > 1:
            parts = dist.all_gather_object(
                [d.tolist() if d is not None else None for d in distinct]
            )
            merged: List[Optional[np.ndarray]] = []
            for j in range(F):
                vals: Optional[set] = set()
                for p in parts:
                    if p[j] is None or vals is None:
                        vals = None
                        break
                    vals.update(p[j])
                    if len(vals) > maxc:
                        vals = None
                        break
                merged.append(
                    np.array(sorted(vals), dtype=np.float64) if vals is not None else None
                )
            distinct = merged
        category_maps: Dict[int, Dict[float, int]] = {}
        for j, d in enumerate(distinct):
            if d is None or len(d) > maxc or len(d) < 1:
                continue
            vals = sorted(float(v) for v in d)
            # 0.0 keeps index 0 (sparsity-preserving, per Spark)
            if 0.0 in vals:
                vals = [0.0] + [v for v in vals if v != 0.0]
            category_maps[j] = {v: i for i, v in enumerate(vals)}
        model = VectorIndexerModel(numFeatures=F, categoryMaps=category_maps)
        self._copyValues(model)
        model._set_parent(self)
        return model


class VectorIndexerModel(_VectorIndexerParams, Model):
    def __init__(self, numFeatures: int = 0,
                 categoryMaps: Optional[Dict[int, Dict[float, int]]] = None):
        super().__init__()
        self.numFeatures = int(numFeatures)
        self.categoryMaps: Dict[int, Dict[float, int]] = {
            int(k): {float(a): int(b) for a, b in v.items()}
            for k, v in (categoryMaps or {}).items()
        }

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("VectorIndexer input must be a vector column")
        handle = self.getHandleInvalid()
        out = x.to(torch.float32).clone()
        keep = None
        cat_meta: Dict[int, int] = {}
        for j, cmap in self.categoryMaps.items():
            vals = torch.tensor(sorted(cmap.keys()), dtype=torch.float64,
                                device=x.device)
            idxs = torch.tensor([cmap[float(v)] for v in vals],
                                dtype=torch.float32, device=x.device)
            col = x[:, j].to(torch.float64)
            pos = torch.searchsorted(vals, col.contiguous())
            pos_c = pos.clamp_max(len(vals) - 1)
            hit = vals[pos_c] == col
            mapped = idxs[pos_c]
            arity = len(cmap)
            if bool((~hit).any()):
                if handle == "error":
                    bad = col[~hit][:5].cpu().tolist()
                    raise ValueError(
                        f"VectorIndexer found unseen values {bad} in feature {j} "
                        f"(handleInvalid='error')"
                    )
                if handle == "keep":
                    mapped = torch.where(hit, mapped, torch.full_like(mapped, float(arity)))
                    arity += 1
                elif handle == "skip":
                    keep = hit if keep is None else (keep & hit)
                else:
                    raise ValueError(f"Unsupported handleInvalid: {handle}")
            out[:, j] = mapped
            cat_meta[j] = arity
        res = dataset.withColumn(self.getOutputCol(), out,
                                 meta={"categorical": cat_meta} if cat_meta else None)
        if keep is not None:
            res = res.filter(keep)
        return res

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# StandardScaler
# ---------------------------------------------------------------------------


class _StandardScalerParams(HasInputCol, HasOutputCol):
    withMean: Param[bool] = Param(
        "undefined", "withMean", "Center data with mean", TypeConverters.toBoolean
    )
    withStd: Param[bool] = Param(
        "undefined", "withStd", "Scale to unit standard deviation", TypeConverters.toBoolean
    )

    def __init__(self):
        super().__init__()
        self._setDefault(withMean=False, withStd=True)

    def getWithMean(self) -> bool:
        return self.getOrDefault(self.withMean)

    def getWithStd(self) -> bool:
        return self.getOrDefault(self.withStd)


class StandardScaler(_StandardScalerParams, Estimator):
    """μ/σ column scaling, withMean=False default (spark:.../feature/StandardScaler.scala)."""

    def __init__(
        self,
        *,
        inputCol: Optional[str] = None,
        outputCol: Optional[str] = None,
        withMean: Optional[bool] = None,
        withStd: Optional[bool] = None,
    ):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol, withMean=withMean, withStd=withStd)

    def _fit(self, dataset: FlowFrame) -> "StandardScalerModel":
        col = dataset.col(self.getInputCol())
        if not (isinstance(col, torch.Tensor) and col.dim() == 2):
            raise TypeError("StandardScaler input must be a vector column")
        from flowml import dist

        x = col.to(torch.float64)
        if dist.get_world_size() > 1:
            # all-reduce (n, Σx, Σx²) moments so sharded fits agree (C6)
            F = x.shape[1]
            buf = torch.zeros(1 + 2 * F, dtype=torch.float64, device=x.device)
            buf[0] = x.shape[0]
            buf[1 : 1 + F] = x.sum(dim=0)
            buf[1 + F :] = (x * x).sum(dim=0)
            dist.all_reduce_(buf)
            n = float(buf[0])
            mean = buf[1 : 1 + F] / max(n, 1.0)
            if n > 1:
                var = (buf[1 + F :] - n * mean * mean) / (n - 1)
                std = var.clamp_min(0).sqrt()
            else:
                std = torch.zeros_like(mean)
        else:
            mean = x.mean(dim=0)
            n = x.shape[0]
            if n > 1:
                std = x.std(dim=0, unbiased=True)  # Spark Summarizer: sample std
            else:
                std = torch.zeros_like(mean)
        model = StandardScalerModel(mean=mean.cpu(), std=std.cpu())
        self._copyValues(model)
        model._set_parent(self)
        return model


class StandardScalerModel(_StandardScalerParams, Model):
    def __init__(self, mean: Optional[torch.Tensor] = None, std: Optional[torch.Tensor] = None):
        super().__init__()
        self.mean = mean
        self.std = std

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        col = dataset.col(self.getInputCol())
        x = col.to(torch.float32)
        # persist the device migration: repeat transforms (and hipGraph
        # capture after one warm-up) see resident tensors, no H2D per call
        self.mean = self.mean.to(device=x.device, dtype=x.dtype)
        self.std = self.std.to(device=x.device, dtype=x.dtype)
        mean = self.mean
        std = self.std
        if self.getWithMean():
            x = x - mean
        if self.getWithStd():
            factor = torch.where(std != 0, 1.0 / std, torch.zeros_like(std))
            x = x * factor
        return dataset.withColumn(self.getOutputCol(), x)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# MinMaxScaler
# ---------------------------------------------------------------------------


class _MinMaxScalerParams(HasInputCol, HasOutputCol):
    min: Param[float] = Param(
        "undefined", "min", "Lower bound after transformation", TypeConverters.toFloat
    )
    max: Param[float] = Param(
        "undefined", "max", "Upper bound after transformation", TypeConverters.toFloat
    )

    def __init__(self):
        super().__init__()
        self._setDefault(min=0.0, max=1.0)

    def getMin(self) -> float:
        return self.getOrDefault(self.min)

    def getMax(self) -> float:
        return self.getOrDefault(self.max)


class MinMaxScaler(_MinMaxScalerParams, Estimator):
    def __init__(
        self,
        *,
        inputCol: Optional[str] = None,
        outputCol: Optional[str] = None,
        min: Optional[float] = None,
        max: Optional[float] = None,
    ):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol, min=min, max=max)

    def _fit(self, dataset: FlowFrame) -> "MinMaxScalerModel":
        col = dataset.col(self.getInputCol())
        if not (isinstance(col, torch.Tensor) and col.dim() == 2):
            raise TypeError("MinMaxScaler input must be a vector column")
        from flowml import dist

        x = col.to(torch.float64)
        mn = dist.all_reduce_(x.min(dim=0).values.contiguous(), op="min")
        mx = dist.all_reduce_(x.max(dim=0).values.contiguous(), op="max")
        model = MinMaxScalerModel(originalMin=mn.cpu(), originalMax=mx.cpu())
        self._copyValues(model)
        model._set_parent(self)
        return model


class MinMaxScalerModel(_MinMaxScalerParams, Model):
    def __init__(
        self,
        originalMin: Optional[torch.Tensor] = None,
        originalMax: Optional[torch.Tensor] = None,
    ):
        super().__init__()
        self.originalMin = originalMin
        self.originalMax = originalMax

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        col = dataset.col(self.getInputCol())
        x = col.to(torch.float32)
        self.originalMin = self.originalMin.to(device=x.device, dtype=x.dtype)
        self.originalMax = self.originalMax.to(device=x.device, dtype=x.dtype)
        omin = self.originalMin
        omax = self.originalMax
        lo, hi = self.getMin(), self.getMax()
        rng = omax - omin
        const = rng == 0
        scale = torch.where(const, torch.zeros_like(rng), (hi - lo) / torch.where(const, torch.ones_like(rng), rng))
        scaled = (x - omin) * scale + lo
        # constant feature → midpoint (Spark behavior)
        mid = torch.full_like(scaled, 0.5 * (hi + lo))
        scaled = torch.where(const.expand_as(scaled), mid, scaled)
        return dataset.withColumn(self.getOutputCol(), scaled)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# IndexToString (maps prediction indices back to label strings)
# ---------------------------------------------------------------------------


class IndexToString(HasInputCol, HasOutputCol, Transformer):
    """Inverse of StringIndexer (spark:.../feature/StringIndexerModel — the
    companion IndexToString transformer): maps a double index column back to
    the original label strings via ``labels``."""

    labels: Param[List[str]] = Param(
        "undefined", "labels", "Optional array of labels specifying index-string mapping.",
        TypeConverters.toListString,
    )

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None,
                 labels: Optional[List[str]] = None):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol, labels=labels)

    def getLabels(self) -> List[str]:
        return self.getOrDefault(self.labels)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        col = dataset.col(self.getInputCol())
        if isinstance(col, np.ndarray):
            raise TypeError("IndexToString input must be a numeric index column")
        if self.isSet(self.labels):
            labels = self.getLabels()
        else:
            # Spark: with no labels param, read them from the input column's
            # ML attributes (StringIndexerModel attaches them)
            labels = dataset.meta(self.getInputCol()).get("labels")
            if not labels:
                raise ValueError(
                    f"IndexToString: no labels set and column "
                    f"{self.getInputCol()!r} carries no label metadata"
                )
        idx = col.detach().cpu().numpy().astype(np.int64)
        if idx.size and (idx.min() < 0 or idx.max() >= len(labels)):
            raise ValueError(
                f"index out of range [0,{len(labels)}) in {self.getInputCol()!r}"
            )
        out = np.asarray(labels, dtype=object)[idx]
        return dataset.withColumn(self.getOutputCol(), out)


# ---------------------------------------------------------------------------
# Normalizer (per-row p-norm scaling)
# ---------------------------------------------------------------------------


class Normalizer(HasInputCol, HasOutputCol, Transformer):
    """Normalize each feature row to unit p-norm (spark:.../feature/Normalizer.scala,
    default p=2)."""

    p: Param[float] = Param("undefined", "p", "the p norm value.", TypeConverters.toFloat)

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None,
                 p: Optional[float] = None):
        super().__init__()
        self._setDefault(p=2.0)
        self._set(inputCol=inputCol, outputCol=outputCol, p=p)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        if not (isinstance(x, torch.Tensor) and x.dim() == 2):
            raise TypeError("Normalizer input must be a vector column")
        pv = self.getOrDefault(self.p)
        norm = torch.norm(x.to(torch.float32), p=pv, dim=1, keepdim=True)
        out = torch.where(norm > 0, x / norm, x)
        return dataset.withColumn(self.getOutputCol(), out)


# ---------------------------------------------------------------------------
# Binarizer (threshold a numeric/vector column to 0/1)
# ---------------------------------------------------------------------------


class Binarizer(HasInputCol, HasOutputCol, Transformer):
    """Binarize continuous features by threshold (spark:.../feature/Binarizer.scala:
    strictly greater than threshold -> 1.0)."""

    threshold: Param[float] = Param(
        "undefined", "threshold", "threshold used to binarize continuous features",
        TypeConverters.toFloat,
    )

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None,
                 threshold: Optional[float] = None):
        super().__init__()
        self._setDefault(threshold=0.0)
        self._set(inputCol=inputCol, outputCol=outputCol, threshold=threshold)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        if isinstance(x, np.ndarray):
            raise TypeError("Binarizer input must be numeric")
        thr = self.getOrDefault(self.threshold)
        out = (x > thr).to(torch.float64 if x.dim() == 1 else torch.float32)
        return dataset.withColumn(self.getOutputCol(), out)


# ---------------------------------------------------------------------------
# MaxAbsScaler (spark:.../feature/MaxAbsScaler.scala)
# ---------------------------------------------------------------------------


class _MaxAbsScalerParams(HasInputCol, HasOutputCol):
    pass


class MaxAbsScaler(_MaxAbsScalerParams, Estimator):
    """Scale each feature to [-1, 1] by dividing by its max absolute value
    (zero-max features pass through unchanged, like Spark)."""

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol)

    def _fit(self, dataset: FlowFrame) -> "MaxAbsScalerModel":
        x = dataset.col(self.getInputCol())
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("MaxAbsScaler input must be a vector column")
        from flowml import dist

        max_abs = dist.all_reduce_(
            x.abs().amax(dim=0).to(torch.float32).contiguous(), op="max"
        ).cpu()
        model = MaxAbsScalerModel(maxAbs=max_abs)
        self._copyValues(model)
        model._set_parent(self)
        return model


class MaxAbsScalerModel(_MaxAbsScalerParams, Model):
    def __init__(self, maxAbs: Optional[torch.Tensor] = None):
        super().__init__()
        self.maxAbs = maxAbs if maxAbs is not None else torch.zeros(0)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        self.maxAbs = self.maxAbs.to(device=x.device, dtype=x.dtype)
        d = torch.where(self.maxAbs == 0, torch.ones_like(self.maxAbs), self.maxAbs)
        return dataset.withColumn(self.getOutputCol(), x / d)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# RobustScaler (spark:.../feature/RobustScaler.scala)
# ---------------------------------------------------------------------------


class _RobustScalerParams(HasInputCol, HasOutputCol):
    lower: Param[float] = Param(
        "undefined", "lower", "Lower quantile to calculate quantile range",
        TypeConverters.toFloat)
    upper: Param[float] = Param(
        "undefined", "upper", "Upper quantile to calculate quantile range",
        TypeConverters.toFloat)
    withCentering: Param[bool] = Param(
        "undefined", "withCentering", "Whether to center data with median",
        TypeConverters.toBoolean)
    withScaling: Param[bool] = Param(
        "undefined", "withScaling",
        "Whether to scale the data to quantile range", TypeConverters.toBoolean)

    def __init__(self):
        super().__init__()
        self._setDefault(lower=0.25, upper=0.75, withCentering=False,
                         withScaling=True)


class RobustScaler(_RobustScalerParams, Estimator):
    """Scale features by subtracting the median and dividing by the
    [lower, upper] quantile range — robust to the heavy-tailed feature
    distributions of flow records (spark:.../feature/RobustScaler.scala:
    defaults lower=0.25, upper=0.75, withCentering=False, withScaling=True).
    Distributed fits use exact type-1 global quantiles over the sharded
    column (``flowml.dist.dist_quantiles``; Spark uses approxQuantile).

    >>> import torch
    >>> from flowml.frame import FlowFrame
    >>> X = torch.tensor([[float(v)] for v in range(1, 10)])
    >>> m = RobustScaler(inputCol="f", outputCol="s").fit(FlowFrame({"f": X}))
    >>> (float(m.median[0]), float(m.range[0]))
    (5.0, 4.0)
    """

    def __init__(self, *, inputCol: Optional[str] = None,
                 outputCol: Optional[str] = None, lower: Optional[float] = None,
                 upper: Optional[float] = None,
                 withCentering: Optional[bool] = None,
                 withScaling: Optional[bool] = None):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol, lower=lower,
                  upper=upper, withCentering=withCentering,
                  withScaling=withScaling)

    def _fit(self, dataset: FlowFrame) -> "RobustScalerModel":
        from flowml import dist

        x = dataset.col(self.getInputCol())
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("RobustScaler input must be a vector column")
        lo = self.getOrDefault(self.lower)
        up = self.getOrDefault(self.upper)
        if not (0 <= lo < up <= 1):
            raise ValueError("require 0 <= lower < upper <= 1")
        xd = x.to(torch.float64)
        F = xd.shape[1]
        if dist.get_world_size() > 1:
            med = torch.empty(F, dtype=torch.float64)
            rng = torch.empty(F, dtype=torch.float64)
            for j in range(F):
                q = dist.dist_quantiles(xd[:, j], [lo, 0.5, up])
                med[j] = q[1]
                rng[j] = q[2] - q[0]
        else:
            qs = torch.quantile(
                xd, torch.tensor([lo, 0.5, up], dtype=torch.float64,
                                 device=xd.device), dim=0)
            med = qs[1].cpu()
            rng = (qs[2] - qs[0]).cpu()
        model = RobustScalerModel(median=med.to(torch.float64),
                                  range_=rng.to(torch.float64))
        self._copyValues(model)
        model._set_parent(self)
        return model


class RobustScalerModel(_RobustScalerParams, Model):
    def __init__(self, median: Optional[torch.Tensor] = None,
                 range_: Optional[torch.Tensor] = None):
        super().__init__()
        self.median = median if median is not None else torch.zeros(0)
        self.range = range_ if range_ is not None else torch.zeros(0)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        med = self.median.to(device=x.device, dtype=x.dtype)
        rng = self.range.to(device=x.device, dtype=x.dtype)
        out = x
        if self.getOrDefault(self.withCentering):
            out = out - med
        if self.getOrDefault(self.withScaling):
            d = torch.where(rng == 0, torch.ones_like(rng), rng)
            out = torch.where(rng == 0, torch.zeros_like(out), out / d)
        return dataset.withColumn(self.getOutputCol(), out)


# ---------------------------------------------------------------------------
# ElementwiseProduct / PolynomialExpansion (spark:.../feature/)
# ---------------------------------------------------------------------------


class ElementwiseProduct(HasInputCol, HasOutputCol, Transformer):
    """Multiply each feature vector by a fixed scaling vector, element-wise
    (spark:.../feature/ElementwiseProduct.scala)."""

    scalingVec: Param[list] = Param(
        "undefined", "scalingVec", "vector for hadamard product",
        TypeConverters.toListFloat)

    def __init__(self, *, inputCol: Optional[str] = None,
                 outputCol: Optional[str] = None,
                 scalingVec: Optional[List[float]] = None):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol, scalingVec=scalingVec)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        w = torch.tensor(self.getOrDefault(self.scalingVec), dtype=x.dtype,
                         device=x.device)
        if x.shape[1] != w.numel():
            raise ValueError(
                f"scalingVec has {w.numel()} entries, features have {x.shape[1]}")
        return dataset.withColumn(self.getOutputCol(), x * w)


class PolynomialExpansion(HasInputCol, HasOutputCol, Transformer):
    """Expand features into a polynomial space of the given degree
    (spark:.../feature/PolynomialExpansion.scala). Output order matches
    Spark's recursive expansion: for input (x, y) at degree 2 the output is
    (x, x², y, xy, y²).

    >>> import torch
    >>> from flowml.frame import FlowFrame
    >>> df = FlowFrame({"f": torch.tensor([[2.0, 3.0]])})
    >>> PolynomialExpansion(inputCol="f", outputCol="p", degree=2
    ...     ).transform(df)["p"].tolist()
    [[2.0, 4.0, 3.0, 6.0, 9.0]]
    """

    degree: Param[int] = Param(
        "undefined", "degree", "the polynomial degree to expand (>= 1)",
        TypeConverters.toInt)

    def __init__(self, *, inputCol: Optional[str] = None,
                 outputCol: Optional[str] = None, degree: Optional[int] = None):
        super().__init__()
        self._setDefault(degree=2)
        self._set(inputCol=inputCol, outputCol=outputCol, degree=degree)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        deg = self.getOrDefault(self.degree)
        if deg < 1:
            raise ValueError("degree must be >= 1")
        N, F = x.shape
        # Spark's order: E(k, d) = ⊕_{p=0..d} v_k^p · E(k−1, d−p) with
        # E(0, d) = [1]; the global constant term is dropped. For (x, y) at
        # degree 3 this yields x, x², x³, y, xy, x²y, y², xy², y³ — the
        # documented PolynomialExpansion sequence.
        def terms(k: int, d: int):
            if k == 0:
                return [()]
            out = []
            for p in range(d + 1):
                for t in terms(k - 1, d - p):
                    out.append(t + (p,))
            return out

        cols: List[torch.Tensor] = []
        for expo in terms(F, deg)[1:]:  # drop the constant
            c = torch.ones(N, dtype=x.dtype, device=x.device)
            for j, e in enumerate(expo):
                if e:
                    c = c * x[:, j] ** e
            cols.append(c)
        return dataset.withColumn(self.getOutputCol(), torch.stack(cols, dim=1))


# ---------------------------------------------------------------------------
# Bucketizer (spark:.../feature/Bucketizer.scala)
# ---------------------------------------------------------------------------


class Bucketizer(HasInputCol, HasOutputCol, HasHandleInvalid, Transformer):
    """Map a continuous column to bucket indices given split points.

    ``splits`` = [s0 < s1 < ... < sn]; bucket i covers [s_i, s_{i+1}) and the
    last bucket is closed on the right. Values outside [s0, sn] (or NaN):
    ``handleInvalid`` error (default) / skip (drop rows) / keep (extra bucket
    index = numBuckets).
    """

    splits: Param[list] = Param(
        "undefined", "splits",
        "Split points for mapping continuous features into buckets (length >= 3, "
        "strictly increasing; -inf/inf cover all values).",
        TypeConverters.toListFloat,
    )

    def __init__(self, *, splits: Optional[list] = None, inputCol: Optional[str] = None,
                 outputCol: Optional[str] = None, handleInvalid: Optional[str] = None):
        super().__init__()
        self._setDefault(handleInvalid="error")
        self._set(splits=splits, inputCol=inputCol, outputCol=outputCol,
                  handleInvalid=handleInvalid)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        sp = self.getOrDefault(self.splits)
        if sp is None or len(sp) < 3:
            raise ValueError("Bucketizer requires splits with length >= 3")
        x = dataset.col(self.getInputCol())
        if isinstance(x, np.ndarray):
            raise TypeError("Bucketizer input must be numeric")
        handle = self.getHandleInvalid()
        edges = torch.tensor(sp, dtype=torch.float64, device=x.device)
        xv = x.reshape(-1).to(torch.float64)
        n_buckets = len(sp) - 1
        idx = torch.bucketize(xv, edges, right=True) - 1  # [s_i, s_{i+1})
        idx = torch.where(xv == edges[-1], torch.full_like(idx, n_buckets - 1), idx)
        invalid = torch.isnan(xv) | (idx < 0) | (idx >= n_buckets)
        if bool(invalid.any()):
            if handle == "error":
                raise ValueError(
                    "Bucketizer found values outside the splits range "
                    "(handleInvalid='error')."
                )
            if handle == "keep":
                idx = torch.where(invalid, torch.full_like(idx, n_buckets), idx)
            elif handle == "skip":
                out = dataset.withColumn(self.getOutputCol(), idx.to(torch.float64))
                return out._take_mask((~invalid).cpu().numpy())
            else:
                raise ValueError(f"Unsupported handleInvalid: {handle}")
        return dataset.withColumn(self.getOutputCol(), idx.to(torch.float64))


# ---------------------------------------------------------------------------
# Imputer (spark:.../feature/Imputer.scala)
# ---------------------------------------------------------------------------


class _ImputerParams(HasInputCol, HasInputCols, HasOutputCol, HasOutputCols):
    strategy: Param[str] = Param(
        "undefined", "strategy",
        "strategy for imputation: mean (default) | median | mode",
        TypeConverters.toString,
    )
    missingValue: Param[float] = Param(
        "undefined", "missingValue",
        "placeholder for missing values (default NaN)",
        TypeConverters.toFloat,
    )
    relativeError: Param[float] = Param(
        "undefined", "relativeError",
        "relative error for the approximate quantile algorithm (API parity; "
        "the median here is computed exactly, so this is accepted and "
        "unused)", TypeConverters.toFloat,
    )

    def _in_out_cols(self):
        if self.isSet(self.inputCols):
            return self.getOrDefault(self.inputCols), self.getOrDefault(self.outputCols)
        return [self.getInputCol()], [self.getOutputCol()]


class Imputer(_ImputerParams, Estimator):
    """Replace missing values with the column mean/median/mode."""

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None,
                 inputCols: Optional[list] = None, outputCols: Optional[list] = None,
                 strategy: Optional[str] = None, missingValue: Optional[float] = None,
                 relativeError: Optional[float] = None):
        super().__init__()
        self._setDefault(strategy="mean", missingValue=float("nan"),
                         relativeError=0.001)
        self._set(inputCol=inputCol, outputCol=outputCol, inputCols=inputCols,
                  outputCols=outputCols, strategy=strategy, missingValue=missingValue, relativeError=relativeError)

    def _fit(self, dataset: FlowFrame) -> "ImputerModel":
        from flowml import dist

        in_cols, _ = self._in_out_cols()
        strat = self.getOrDefault(self.strategy)
        mv = self.getOrDefault(self.missingValue)
        world = dist.get_world_size()
        surrogates = []
        for c in in_cols:
            x = dataset.col(c).reshape(-1).to(torch.float64)
            miss = torch.isnan(x) if math.isnan(mv) else (x == mv)
            good = x[~miss & ~torch.isnan(x)]
            n_good = torch.tensor([float(len(good))], dtype=torch.float64, device=x.device)
            if world > 1:
                dist.all_reduce_(n_good)
            if float(n_good) == 0:
                raise ValueError(f"Imputer: column {c!r} has no non-missing values")
            if strat == "mean":
                s = torch.zeros(1, dtype=torch.float64, device=x.device)
                if len(good):
                    s[0] = good.sum()
                if world > 1:
                    dist.all_reduce_(s)
                surrogates.append(float(s / n_good))
            elif strat == "median":
                if world > 1:
                    surrogates.append(float(dist.dist_quantiles(good, [0.5])[0]))
                else:
                    surrogates.append(float(good.median()))
            elif strat == "mode":
                vals, cnts = torch.unique(good, return_counts=True)
                if world > 1:
                    merged: Dict[float, int] = {}
                    for part in dist.all_gather_object(
                        dict(zip((float(v) for v in vals), (int(v) for v in cnts)))
                    ):
                        for k, v in part.items():
                            merged[k] = merged.get(k, 0) + v
                    # ties: smallest value wins (deterministic across ranks)
                    best = max(sorted(merged), key=lambda k: merged[k])
                    surrogates.append(float(best))
                else:
                    surrogates.append(float(vals[int(cnts.argmax())]))
            else:
                raise ValueError(f"Unsupported strategy {strat!r}")
        model = ImputerModel(surrogates=surrogates)
        self._copyValues(model)
        model._set_parent(self)
        return model


class ImputerModel(_ImputerParams, Model):
    def __init__(self, surrogates: Optional[list] = None):
        super().__init__()
        self.surrogates = list(surrogates or [])

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        in_cols, out_cols = self._in_out_cols()
        mv = self.getOrDefault(self.missingValue)
        out = dataset
        for c, o, sur in zip(in_cols, out_cols, self.surrogates):
            x = out.col(c)
            miss = torch.isnan(x) if math.isnan(mv) else (x == mv)
            out = out.withColumn(o, torch.where(miss, torch.full_like(x, sur), x))
        return out

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# QuantileDiscretizer (spark:.../feature/QuantileDiscretizer.scala)
# ---------------------------------------------------------------------------


class QuantileDiscretizer(HasInputCol, HasOutputCol, HasHandleInvalid, Estimator):
    """Fit a Bucketizer whose splits are approximate quantiles of the input
    column (numBuckets equi-depth buckets, -inf/+inf end splits, duplicate
    quantiles collapsed — so the fitted model can have fewer buckets)."""

    numBuckets: Param[int] = Param(
        "undefined", "numBuckets",
        "Number of buckets (quantiles) into which data points are grouped (>= 2).",
        TypeConverters.toInt,
    )
    relativeError: Param[float] = Param(
        "undefined", "relativeError",
        "relative error for the approximate quantile algorithm (API parity; "
        "quantiles here are exact — torch.quantile locally, type-1 bisection "
        "distributed — so this is accepted and unused)",
        TypeConverters.toFloat,
    )

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None,
                 numBuckets: Optional[int] = None, handleInvalid: Optional[str] = None,
                 relativeError: Optional[float] = None):
        super().__init__()
        self._setDefault(numBuckets=2, handleInvalid="error", relativeError=0.001)
        self._set(inputCol=inputCol, outputCol=outputCol, numBuckets=numBuckets,
                  handleInvalid=handleInvalid, relativeError=relativeError)

    def _fit(self, dataset: FlowFrame) -> "Bucketizer":
        x = dataset.col(self.getInputCol())
        if isinstance(x, np.ndarray):
            raise TypeError("QuantileDiscretizer input must be numeric")
        nb = self.getOrDefault(self.numBuckets)
        if nb < 2:
            raise ValueError("numBuckets must be >= 2")
        from flowml import dist

        xv = x.reshape(-1).to(torch.float64)
        xv = xv[~torch.isnan(xv)]
        q = torch.arange(1, nb, dtype=torch.float64, device=xv.device) / nb
        if dist.get_world_size() > 1:
            # sharded data: global type-1 quantiles (Spark itself uses
            # approxQuantile here, so the interpolation difference vs the
            # local torch.quantile path is within the contract)
            inner = dist.dist_quantiles(xv, [float(p) for p in q])
        else:
            inner = torch.quantile(xv, q) if len(xv) else torch.zeros(0, dtype=torch.float64)
        vals = sorted(set(float(v) for v in inner))
        splits = [-float("inf")] + vals + [float("inf")]
        return Bucketizer(splits=splits, inputCol=self.getInputCol(),
                          outputCol=self.getOutputCol(),
                          handleInvalid=self.getHandleInvalid())


# ---------------------------------------------------------------------------
# RFormula (spark:.../feature/RFormula.scala)
# ---------------------------------------------------------------------------


class _RFormulaParams(Params):
    formula: Param[str] = Param(
        "undefined", "formula", "R model formula", TypeConverters.toString)
    featuresCol: Param[str] = Param(
        "undefined", "featuresCol", "features column name", TypeConverters.toString)
    labelCol: Param[str] = Param(
        "undefined", "labelCol", "label column name", TypeConverters.toString)
    handleInvalid: Param[str] = Param(
        "undefined", "handleInvalid",
        "how to handle invalid entries: error|skip|keep", TypeConverters.toString)

    def __init__(self):
        super().__init__()
        self._setDefault(featuresCol="features", labelCol="label",
                         handleInvalid="error")


def _parse_r_formula(formula: str, columns: List[str]):
    """Parse ``label ~ term + term - term`` with ``.`` = all other columns
    and ``a:b`` numeric interactions. Returns (label, terms) where each term
    is a single column name or a tuple of names (interaction)."""
    if "~" not in formula:
        raise ValueError(f"formula must contain '~': {formula!r}")
    lhs, rhs = formula.split("~", 1)
    label = lhs.strip()
    if not label:
        raise ValueError("formula needs a label on the left of '~'")
    include: List = []
    exclude: set = set()
    for chunk in rhs.replace("-", "+-").split("+"):
        t = chunk.strip()
        if not t:
            continue
        neg = t.startswith("-")
        if neg:
            t = t[1:].strip()
        if t == ".":
            for c in columns:
                if c != label and c not in include:
                    include.append(c)
        elif ":" in t:
            parts = tuple(p.strip() for p in t.split(":"))
            (exclude.add(parts) if neg else include.append(parts))
        else:
            (exclude.add(t) if neg else include.append(t))
    terms = [t for t in include if t not in exclude]
    return label, terms


class RFormula(_RFormulaParams, Estimator):
    """R-style model formula (spark:.../feature/RFormula.scala): string
    columns are StringIndexed + one-hot encoded (dropLast), numeric columns
    pass through, ``a:b`` numeric interactions multiply, ``.`` expands to
    all non-label columns and ``- x`` excludes; a string label is
    StringIndexed into ``labelCol``.

    >>> import numpy as np, torch
    >>> from flowml.frame import FlowFrame
    >>> df = FlowFrame({"clicked": np.array(["no", "yes", "no"], dtype=object),
    ...                 "country": np.array(["us", "ca", "us"], dtype=object),
    ...                 "hour": torch.tensor([18.0, 12.0, 15.0])})
    >>> m = RFormula(formula="clicked ~ country + hour").fit(df)
    >>> out = m.transform(df)
    >>> out["features"].shape
    torch.Size([3, 2])
    >>> out["label"].tolist()
    [0.0, 1.0, 0.0]
    """

    def __init__(self, *, formula: Optional[str] = None,
                 featuresCol: Optional[str] = None,
                 labelCol: Optional[str] = None,
                 handleInvalid: Optional[str] = None):
        super().__init__()
        self._set(formula=formula, featuresCol=featuresCol, labelCol=labelCol,
                  handleInvalid=handleInvalid)

    def setFormula(self, v: str):
        return self.set(self.formula, v)

    def _fit(self, dataset: FlowFrame) -> "RFormulaModel":
        label, terms = _parse_r_formula(self.getOrDefault(self.formula),
                                        dataset.columns)
        handle = self.getOrDefault(self.handleInvalid)
        stages: List = []
        assemble: List[str] = []
        tmp = dataset
        for t in terms:
            if isinstance(t, tuple):
                for c in t:
                    if tmp.is_string(c):
                        raise ValueError(
                            f"interaction term {':'.join(t)} requires numeric "
                            f"columns; {c!r} is a string column")
                assemble.append("__rf_inter_" + "_".join(t))
                continue
            if tmp.is_string(t):
                si = StringIndexer(inputCol=t, outputCol=f"__rf_{t}_idx",
                                   handleInvalid="keep" if handle == "keep" else handle)
                sim = si.fit(tmp)
                tmp = sim.transform(tmp)
                ohe = OneHotEncoder(inputCol=f"__rf_{t}_idx",
                                    outputCol=f"__rf_{t}_vec",
                                    handleInvalid="keep" if handle == "keep" else "error")
                ohm = ohe.fit(tmp)
                tmp = ohm.transform(tmp)
                stages += [sim, ohm]
                assemble.append(f"__rf_{t}_vec")
            else:
                assemble.append(t)
        label_stage = None
        if label in dataset.columns and dataset.is_string(label):
            label_stage = StringIndexer(
                inputCol=label, outputCol=self.getOrDefault(self.labelCol)
            ).fit(dataset)
        model = RFormulaModel(
            terms=terms, stages=stages, assemble_cols=assemble,
            label_source=label, label_stage=label_stage)
        self._copyValues(model)
        model._set_parent(self)
        return model


class RFormulaModel(_RFormulaParams, Model):
    def __init__(self, terms=None, stages=None, assemble_cols=None,
                 label_source: str = "", label_stage=None):
        super().__init__()
        self._terms = terms or []
        self._stages = stages or []
        self._assemble = assemble_cols or []
        self._label_source = label_source
        self._label_stage = label_stage

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        out = dataset
        for st in self._stages:
            out = st.transform(out)
        for t in self._terms:
            if isinstance(t, tuple):
                col = out.col(t[0]).reshape(-1).clone()
                for c in t[1:]:
                    col = col * out.col(c).reshape(-1)
                out = out.withColumn("__rf_inter_" + "_".join(t), col)
        out = VectorAssembler(
            inputCols=self._assemble,
            outputCol=self.getOrDefault(self.featuresCol),
            handleInvalid=self.getOrDefault(self.handleInvalid),
        ).transform(out)
        lab = self.getOrDefault(self.labelCol)
        if self._label_stage is not None:
            out = self._label_stage.transform(out)
        elif self._label_source in out.columns and self._label_source != lab:
            out = out.withColumn(lab, out.col(self._label_source))
        # drop the internal temp columns
        drop = [c for c in out.columns if c.startswith("__rf_")]
        return out.drop(*drop) if drop else out


# ---------------------------------------------------------------------------
# UnivariateFeatureSelector / ChiSqSelector
# (spark:.../feature/UnivariateFeatureSelector.scala, ChiSqSelector.scala)
# ---------------------------------------------------------------------------


def _bh_fdr_select(pvals: np.ndarray, alpha: float) -> np.ndarray:
    """Benjamini-Hochberg: indices passing FDR control at level alpha."""
    F = len(pvals)
    order = np.argsort(pvals, kind="stable")
    ranked = pvals[order]
    thresh = alpha * (np.arange(1, F + 1) / F)
    ok = ranked <= thresh
    if not ok.any():
        return np.zeros(0, dtype=np.int64)
    kmax = int(np.nonzero(ok)[0].max())
    return np.sort(order[: kmax + 1])


def _selector_stats(X: torch.Tensor, y: torch.Tensor, feature_type: str,
                    label_type: str):
    """Per-feature (score, p-value) with all-reduced sufficient statistics.

    chi2 for categorical/categorical, one-way ANOVA F (f_classif) for
    continuous/categorical, univariate regression F (f_regression) for
    continuous/continuous — the three combinations Spark supports.
    """
    from scipy import stats as sstats

    from flowml import dist

    Xd = X.to(torch.float64)
    yd = y.reshape(-1).to(torch.float64)
    N, F = Xd.shape
    if feature_type == "categorical" and label_type == "categorical":
        xa = Xd.cpu().numpy().astype(np.int64)
        ya = yd.cpu().numpy().astype(np.int64)
        if xa.min(initial=0) < 0 or ya.min(initial=0) < 0:
            raise ValueError("categorical values must be non-negative indices")
        vmax = torch.tensor([float(xa.max(initial=0)),
                             float(ya.max(initial=0))], device=X.device)
        dist.all_reduce_(vmax, op="max")
        vmax = vmax.cpu()
        V, C = int(vmax[0]) + 1, int(vmax[1]) + 1
        if F * V * C > 100_000_000:
            raise ValueError(
                f"chi2 contingency would need {F}x{V}x{C} cells — a feature "
                "declared categorical has huge arity; use "
                "featureType='continuous' for continuous columns")
        # contingency counts per feature: F×V×C, flat all_reduce
        cont = np.zeros((F, V, C), np.float64)
        for j in range(F):
            np.add.at(cont[j], (xa[:, j], ya), 1.0)
        # all_reduce on the features' device (NCCL needs device tensors;
        # gloo handles CPU) — then host math
        cont_t = torch.from_numpy(cont.reshape(-1)).to(X.device)
        dist.all_reduce_(cont_t)
        cont = cont_t.cpu().numpy().reshape(F, V, C)
        scores = np.zeros(F)
        pvals = np.ones(F)
        for j in range(F):
            tbl = cont[j]
            rows = tbl.sum(axis=1)
            cols = tbl.sum(axis=0)
            n = tbl.sum()
            keep_r = rows > 0
            keep_c = cols > 0
            t = tbl[np.ix_(keep_r, keep_c)]
            exp = np.outer(rows[keep_r], cols[keep_c]) / max(n, 1.0)
            dof = (t.shape[0] - 1) * (t.shape[1] - 1)
            if dof <= 0:
                continue
            scores[j] = float(((t - exp) ** 2 / exp).sum())
            pvals[j] = float(sstats.chi2.sf(scores[j], dof))
        return scores, pvals
    if feature_type == "continuous" and label_type == "categorical":
        ya = yd.cpu().numpy().astype(np.int64)
        kmax = torch.tensor([float(ya.max(initial=0))], device=Xd.device)
        dist.all_reduce_(kmax, op="max")
        kmax = kmax.cpu()
        K = int(kmax[0]) + 1
        oh = torch.zeros(N, K, dtype=torch.float64, device=Xd.device)
        oh[torch.arange(N, device=Xd.device), yd.to(torch.int64)] = 1.0
        buf = torch.cat([
            oh.sum(dim=0),                     # K class counts
            (oh.T @ Xd).reshape(-1),           # K×F class sums
            (oh.T @ (Xd * Xd)).reshape(-1),    # K×F class sum-squares
        ])
        dist.all_reduce_(buf)                  # on-device (NCCL-safe)
        buf = buf.cpu()
        nk = buf[:K].numpy()
        sk = buf[K:K + K * F].view(K, F).numpy()
        qk = buf[K + K * F:].view(K, F).numpy()
        n = nk.sum()
        valid = nk > 0
        k_eff = int(valid.sum())
        mean_k = sk[valid] / nk[valid, None]
        gmean = sk.sum(axis=0) / n
        ss_between = (nk[valid, None] * (mean_k - gmean) ** 2).sum(axis=0)
        ss_within = (qk[valid] - nk[valid, None] * mean_k ** 2).sum(axis=0)
        df_b = k_eff - 1
        df_w = n - k_eff
        with np.errstate(divide="ignore", invalid="ignore"):
            scores = (ss_between / df_b) / np.maximum(ss_within / df_w, 1e-300)
        pvals = sstats.f.sf(scores, df_b, df_w)
        return scores, pvals
    if feature_type == "continuous" and label_type == "continuous":
        buf = torch.cat([
            torch.tensor([float(N)], dtype=torch.float64, device=Xd.device),
            Xd.sum(dim=0), (Xd * Xd).sum(dim=0),
            (Xd * yd.unsqueeze(1)).sum(dim=0),
            yd.sum().reshape(1), (yd * yd).sum().reshape(1),
        ])
        dist.all_reduce_(buf)                  # on-device (NCCL-safe)
        buf = buf.cpu()
        n = float(buf[0])
        sx = buf[1:1 + F].numpy()
        qx = buf[1 + F:1 + 2 * F].numpy()
        sxy = buf[1 + 2 * F:1 + 3 * F].numpy()
        sy = float(buf[-2])
        qy = float(buf[-1])
        cov = sxy / n - (sx / n) * (sy / n)
        vx = np.maximum(qx / n - (sx / n) ** 2, 1e-300)
        vy = max(qy / n - (sy / n) ** 2, 1e-300)
        r2 = np.clip(cov ** 2 / (vx * vy), 0.0, 1.0 - 1e-15)
        dfn = n - 2
        scores = r2 / (1 - r2) * dfn
        pvals = sstats.f.sf(scores, 1, dfn)
        return scores, pvals
    raise ValueError(
        f"unsupported featureType={feature_type!r} / labelType={label_type!r} "
        "(Spark supports categorical/categorical, continuous/categorical, "
        "continuous/continuous)")


class _UnivariateFeatureSelectorParams(HasInputCol, HasOutputCol, HasLabelCol):
    featureType: Param[str] = Param(
        "undefined", "featureType", "The feature type: categorical|continuous",
        TypeConverters.toString)
    labelType: Param[str] = Param(
        "undefined", "labelType", "The label type: categorical|continuous",
        TypeConverters.toString)
    selectionMode: Param[str] = Param(
        "undefined", "selectionMode",
        "The selection mode: numTopFeatures|percentile|fpr|fdr|fwe",
        TypeConverters.toString)
    selectionThreshold: Param[float] = Param(
        "undefined", "selectionThreshold",
        "The upper bound of the features that selector will select",
        TypeConverters.toFloat)

    def __init__(self):
        super().__init__()
        self._setDefault(labelCol="label", selectionMode="numTopFeatures")


class UnivariateFeatureSelector(_UnivariateFeatureSelectorParams, Estimator):
    """Select features via univariate statistical tests against the label
    (spark:.../feature/UnivariateFeatureSelector.scala): chi2 for
    categorical features + categorical label, ANOVA F-test for continuous +
    categorical, univariate-regression F-test for continuous + continuous.
    Sufficient statistics (contingency counts / per-class moments /
    correlation moments) are all-reduced, so sharded fits agree.
    selectionMode thresholds default per Spark: numTopFeatures 50,
    percentile 0.1, fpr/fdr/fwe 0.05. Selection orders by ascending
    p-value (ties: lower feature index first)."""

    def __init__(self, *, featuresCol: Optional[str] = None,
                 outputCol: Optional[str] = None,
                 labelCol: Optional[str] = None,
                 featureType: Optional[str] = None,
                 labelType: Optional[str] = None,
                 selectionMode: Optional[str] = None,
                 selectionThreshold: Optional[float] = None):
        super().__init__()
        self._set(inputCol=featuresCol, outputCol=outputCol, labelCol=labelCol,
                  featureType=featureType, labelType=labelType,
                  selectionMode=selectionMode,
                  selectionThreshold=selectionThreshold)

    def _fit(self, dataset: FlowFrame) -> "UnivariateFeatureSelectorModel":
        X = dataset.col(self.getOrDefault(self.inputCol))
        y = dataset.col(self.getOrDefault(self.labelCol))
        if not isinstance(X, torch.Tensor) or X.dim() != 2:
            raise TypeError("selector features must be a vector column")
        ftype = self.getOrDefault(self.featureType)
        ltype = self.getOrDefault(self.labelType)
        mode = self.getOrDefault(self.selectionMode)
        if self.isSet(self.selectionThreshold):
            thr = self.getOrDefault(self.selectionThreshold)
        else:
            thr = {"numTopFeatures": 50.0, "percentile": 0.1,
                   "fpr": 0.05, "fdr": 0.05, "fwe": 0.05}[mode]
        scores, pvals = _selector_stats(X, torch.as_tensor(
            y if isinstance(y, torch.Tensor) else np.asarray(y, np.float64)),
            ftype, ltype)
        F = len(pvals)
        order = np.argsort(pvals, kind="stable")
        if mode == "numTopFeatures":
            sel = np.sort(order[: int(thr)])
        elif mode == "percentile":
            sel = np.sort(order[: int(thr * F)])
        elif mode == "fpr":
            sel = np.nonzero(pvals < thr)[0]
        elif mode == "fwe":
            sel = np.nonzero(pvals < thr / F)[0]
        elif mode == "fdr":
            sel = _bh_fdr_select(pvals, thr)
        else:
            raise ValueError(f"Unsupported selectionMode {mode!r}")
        model = UnivariateFeatureSelectorModel(
            selectedFeatures=[int(i) for i in sel])
        self._copyValues(model)
        model._set_parent(self)
        return model


class UnivariateFeatureSelectorModel(_UnivariateFeatureSelectorParams, Model):
    def __init__(self, selectedFeatures: Optional[List[int]] = None):
        super().__init__()
        self.selectedFeatures: List[int] = list(selectedFeatures or [])

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getOrDefault(self.inputCol))
        idx = torch.tensor(self.selectedFeatures, dtype=torch.int64,
                           device=x.device)
        return dataset.withColumn(self.getOrDefault(self.outputCol),
                                  x.index_select(1, idx))

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


class ChiSqSelector(HasInputCol, HasOutputCol, HasLabelCol, Estimator):
    """Legacy chi-squared feature selector (spark:.../feature/
    ChiSqSelector.scala, deprecated upstream in favor of
    UnivariateFeatureSelector): categorical features vs categorical label,
    numTopFeatures default 50."""

    numTopFeatures: Param[int] = Param(
        "undefined", "numTopFeatures",
        "Number of features that selector will select", TypeConverters.toInt)

    def __init__(self, *, featuresCol: Optional[str] = None,
                 outputCol: Optional[str] = None,
                 labelCol: Optional[str] = None,
                 numTopFeatures: Optional[int] = None):
        super().__init__()
        self._setDefault(labelCol="label", numTopFeatures=50)
        self._set(inputCol=featuresCol, outputCol=outputCol, labelCol=labelCol,
                  numTopFeatures=numTopFeatures)

    def _fit(self, dataset: FlowFrame) -> UnivariateFeatureSelectorModel:
        sel = UnivariateFeatureSelector(
            featuresCol=self.getOrDefault(self.inputCol),
            outputCol=self.getOrDefault(self.outputCol),
            labelCol=self.getOrDefault(self.labelCol),
            featureType="categorical", labelType="categorical",
            selectionMode="numTopFeatures",
            selectionThreshold=float(self.getOrDefault(self.numTopFeatures)),
        )
        model = sel._fit(dataset)
        model._set(inputCol=self.getOrDefault(self.inputCol),
                   outputCol=self.getOrDefault(self.outputCol))
        model._set_parent(self)
        return model


# ---------------------------------------------------------------------------
# VarianceThresholdSelector / VectorSizeHint (spark:.../feature/)
# ---------------------------------------------------------------------------


class VarianceThresholdSelector(HasOutputCol, Estimator):
    """Drop features whose (sample) variance is <= varianceThreshold
    (spark:.../feature/VarianceThresholdSelector.scala, default 0 — removes
    constant features). Moments are all-reduced, so sharded fits agree."""

    featuresCol: Param[str] = Param(
        "undefined", "featuresCol", "features column name",
        TypeConverters.toString)
    varianceThreshold: Param[float] = Param(
        "undefined", "varianceThreshold",
        "Param for variance threshold. Features with a (sample) variance "
        "not greater than this threshold will be removed.",
        TypeConverters.toFloat)

    def __init__(self, *, featuresCol: Optional[str] = None,
                 outputCol: Optional[str] = None,
                 varianceThreshold: Optional[float] = None):
        super().__init__()
        self._setDefault(featuresCol="features", varianceThreshold=0.0)
        self._set(featuresCol=featuresCol, outputCol=outputCol,
                  varianceThreshold=varianceThreshold)

    def _fit(self, dataset: FlowFrame) -> "VarianceThresholdSelectorModel":
        from flowml import dist

        x = dataset.col(self.getOrDefault(self.featuresCol))
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("VarianceThresholdSelector input must be a vector column")
        xd = x.to(torch.float64)
        F = xd.shape[1]
        buf = torch.cat([
            torch.tensor([float(xd.shape[0])], dtype=torch.float64,
                         device=xd.device),
            xd.sum(dim=0), (xd * xd).sum(dim=0)])
        dist.all_reduce_(buf)
        n = float(buf[0])
        mean = buf[1:1 + F] / max(n, 1.0)
        var = (buf[1 + F:] - n * mean * mean) / max(n - 1.0, 1.0)  # sample var
        thr = self.getOrDefault(self.varianceThreshold)
        sel = [int(i) for i in torch.nonzero(var.cpu() > thr).flatten()]
        model = VarianceThresholdSelectorModel(selectedFeatures=sel)
        self._copyValues(model)
        model._set_parent(self)
        return model


class VarianceThresholdSelectorModel(HasOutputCol, Model):
    featuresCol: Param[str] = Param(
        "undefined", "featuresCol", "features column name",
        TypeConverters.toString)

    def __init__(self, selectedFeatures: Optional[List[int]] = None):
        super().__init__()
        self._setDefault(featuresCol="features")
        self.selectedFeatures: List[int] = list(selectedFeatures or [])

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getOrDefault(self.featuresCol))
        idx = torch.tensor(self.selectedFeatures, dtype=torch.int64,
                           device=x.device)
        return dataset.withColumn(self.getOrDefault(self.outputCol),
                                  x.index_select(1, idx))

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


class VectorSizeHint(HasInputCol, HasHandleInvalid, Transformer):
    """Validate (and annotate) a vector column's size
    (spark:.../feature/VectorSizeHint.scala). Dense tensors always carry
    their size here, so this validates and stamps {"size": n} metadata."""

    size: Param[int] = Param(
        "undefined", "size", "the size of the vector that this hint is for",
        TypeConverters.toInt)

    def __init__(self, *, inputCol: Optional[str] = None,
                 size: Optional[int] = None,
                 handleInvalid: Optional[str] = None):
        super().__init__()
        self._setDefault(handleInvalid="error")
        self._set(inputCol=inputCol, size=size, handleInvalid=handleInvalid)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        col = self.getInputCol()
        x = dataset.col(col)
        want = self.getOrDefault(self.size)
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("VectorSizeHint input must be a vector column")
        if x.shape[1] != want and self.getHandleInvalid() == "error":
            raise ValueError(
                f"column {col!r} has vector size {x.shape[1]}, expected {want}")
        meta = dataset.meta(col)
        meta["size"] = int(want)
        return dataset.withMeta(col, meta)


# ---------------------------------------------------------------------------
# PCA (spark:.../feature/PCA.scala)
# ---------------------------------------------------------------------------


class _PCAParams(HasInputCol, HasOutputCol):
    k: Param[int] = Param(
        "undefined", "k", "the number of principal components", TypeConverters.toInt
    )


class PCA(_PCAParams, Estimator):
    """Project vectors onto the top-k principal components. The covariance
    moments are per-rank reductions all-reduced across GPUs (SURVEY.md §2.5
    C6); the K×K eigendecomposition runs replicated on every rank."""

    def __init__(self, *, k: Optional[int] = None, inputCol: Optional[str] = None,
                 outputCol: Optional[str] = None):
        super().__init__()
        self._set(k=k, inputCol=inputCol, outputCol=outputCol)

    def _fit(self, dataset: FlowFrame) -> "PCAModel":
        from flowml import dist

        x = dataset.col(self.getInputCol())
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("PCA input must be a vector column")
        k = self.getOrDefault(self.k)
        N, F = x.shape
        if k is None or not (1 <= k <= F):
            raise ValueError(f"k must be in [1, {F}]")
        X = x.to(torch.float64)
        buf = torch.zeros(1 + F + F * F, dtype=torch.float64, device=X.device)
        buf[0] = N
        buf[1 : 1 + F] = X.sum(dim=0)
        buf[1 + F :] = (X.T @ X).reshape(-1)
        dist.all_reduce_(buf)
        n = float(buf[0])
        mean = buf[1 : 1 + F] / n
        cov = (buf[1 + F :].view(F, F) - n * torch.outer(mean, mean)) / max(n - 1, 1.0)
        evals, evecs = torch.linalg.eigh(cov.cpu())  # ascending
        order = torch.argsort(evals, descending=True)[:k]
        comps = evecs[:, order]                       # F×k
        ev = evals[order].clamp_min(0)
        total = float(evals.clamp_min(0).sum())
        model = PCAModel(
            pc=comps.to(torch.float64),
            explainedVariance=(ev / total if total > 0 else ev).to(torch.float64),
        )
        self._copyValues(model)
        model._set_parent(self)
        return model


class PCAModel(_PCAParams, Model):
    def __init__(self, pc: Optional[torch.Tensor] = None,
                 explainedVariance: Optional[torch.Tensor] = None):
        super().__init__()
        self._pc = pc if pc is not None else torch.zeros(0, 0, dtype=torch.float64)
        self._ev = explainedVariance if explainedVariance is not None else torch.zeros(0)

    @property
    def pc(self):
        from flowml.linalg import Matrices

        return Matrices.dense(self._pc.shape[0], self._pc.shape[1],
                              self._pc.numpy().T.flatten())

    @property
    def explainedVariance(self):
        from flowml.linalg import DenseVector

        return DenseVector(self._ev.numpy())

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        pc = self._pc.to(device=x.device, dtype=x.dtype)
        return dataset.withColumn(self.getOutputCol(), x @ pc)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str) -> None:
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)


# ---------------------------------------------------------------------------
# VectorSlicer / Interaction (spark:.../feature/VectorSlicer.scala, Interaction.scala)
# ---------------------------------------------------------------------------


class VectorSlicer(HasInputCol, HasOutputCol, Transformer):
    """Select a sub-vector by indices (Spark's name-based selection is not
    supported — our vector columns carry no attribute metadata)."""

    indices: Param[list] = Param(
        "undefined", "indices",
        "An array of indices to select features from a vector column.",
        TypeConverters.toListInt,
    )

    def __init__(self, *, inputCol: Optional[str] = None, outputCol: Optional[str] = None,
                 indices: Optional[list] = None):
        super().__init__()
        self._set(inputCol=inputCol, outputCol=outputCol, indices=indices)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        x = dataset.col(self.getInputCol())
        if not isinstance(x, torch.Tensor) or x.dim() != 2:
            raise TypeError("VectorSlicer input must be a vector column")
        idx = self.getOrDefault(self.indices)
        if not idx:
            raise ValueError("VectorSlicer requires non-empty indices")
        sel = torch.tensor(idx, dtype=torch.int64, device=x.device)
        return dataset.withColumn(self.getOutputCol(), x.index_select(1, sel))


class Interaction(HasInputCols, HasOutputCol, Transformer):
    """Elementwise product-expansion of the input columns: the output vector
    holds every cross-product combination of one element from each input
    (scalars are width-1 vectors) — Spark's Interaction transformer."""

    def __init__(self, *, inputCols: Optional[list] = None, outputCol: Optional[str] = None):
        super().__init__()
        self._set(inputCols=inputCols, outputCol=outputCol)

    def _transform(self, dataset: FlowFrame) -> FlowFrame:
        cols = self.getInputCols()
        if not cols or len(cols) < 2:
            raise ValueError("Interaction requires at least two inputCols")
        mats = []
        for c in cols:
            t = dataset.col(c)
            if isinstance(t, np.ndarray):
                raise TypeError(f"Interaction input {c!r} must be numeric")
            mats.append((t if t.dim() == 2 else t.reshape(-1, 1)).to(torch.float32))
        out = mats[0]
        for t in mats[1:]:
            # (N, a) x (N, b) -> (N, a*b)
            out = (out.unsqueeze(2) * t.unsqueeze(1)).reshape(out.shape[0], -1)
        return dataset.withColumn(self.getOutputCol(), out)
