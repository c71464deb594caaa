"""User-visible linear algebra types: DenseVector, SparseVector, matrices.

Behavioral spec: ``pyspark.ml.linalg`` (upstream
``spark:python/pyspark/ml/linalg/__init__.py``; SURVEY.md §2.2 "Vectors" row).
These types appear in the public API (model coefficients, assembled feature
rows) and in the saved-model Parquet schemas, so their reprs and round-trip
semantics matter. Backing storage is numpy; bulk columnar data lives in
``flowml.frame`` as torch tensors — these objects are the per-row / per-model
view only.
"""

from __future__ import annotations

from typing import Sequence, Union

import numpy as np


class Vector:
    def toArray(self) -> np.ndarray:
        raise NotImplementedError


class DenseVector(Vector):
    """Dense float64 vector, semantics of pyspark.ml.linalg.DenseVector."""

    def __init__(self, values: Union[Sequence[float], np.ndarray]):
        self.values = np.asarray(values, dtype=np.float64)
        if self.values.ndim != 1:
            raise ValueError("DenseVector requires a 1-D input")

    @property
    def size(self) -> int:
        return len(self.values)

    def toArray(self) -> np.ndarray:
        return self.values

    def dot(self, other) -> float:
        if isinstance(other, SparseVector):
            return float(np.dot(self.values[other.indices], other.values))
        if isinstance(other, DenseVector):
            other = other.values
        return float(np.dot(self.values, np.asarray(other, dtype=np.float64)))

    def norm(self, p) -> float:
        return float(np.linalg.norm(self.values, p))

    def squared_distance(self, other) -> float:
        if isinstance(other, SparseVector):
            return other.squared_distance(self)
        if isinstance(other, DenseVector):
            other = other.values
        diff = self.values - np.asarray(other, dtype=np.float64)
        return float(np.dot(diff, diff))

    def __len__(self) -> int:
        return len(self.values)

    def __getitem__(self, item):
        return self.values[item]

    def __iter__(self):
        return iter(self.values)

    def __eq__(self, other) -> bool:
        if isinstance(other, DenseVector):
            return np.array_equal(self.values, other.values)
        if isinstance(other, SparseVector):
            return len(self) == other.size and np.array_equal(self.values, other.toArray())
        return False

    def __hash__(self):
        size = len(self.values)
        result = 31 + size
        nnz = 0
        i = 0
        while i < size and nnz < 128:
            if self.values[i] != 0:
                result = 31 * result + i
                bits = np.float64(self.values[i]).view(np.int64)
                result = 31 * result + int(bits ^ (bits >> 32))
                nnz += 1
            i += 1
        return int(result)

    def __repr__(self) -> str:
        # pyspark formats with ", " separators
        vals = ", ".join(_format_float(v) for v in self.values)
        return f"DenseVector([{vals}])"

    def __str__(self) -> str:
        return "[" + ",".join(_format_float(v) for v in self.values) + "]"


class SparseVector(Vector):
    """Sparse vector ``(size, indices, values)``, pyspark.ml.linalg semantics."""

    def __init__(self, size: int, *args):
        self.size = int(size)
        if len(args) == 1:
            pairs = args[0]
            if isinstance(pairs, dict):
                pairs = sorted(pairs.items())
            indices, values = zip(*pairs) if pairs else ((), ())
        elif len(args) == 2:
            indices, values = args
        else:
            raise TypeError("SparseVector expects (size, pairs) or (size, indices, values)")
        self.indices = np.asarray(indices, dtype=np.int32)
        self.values = np.asarray(values, dtype=np.float64)
        if self.indices.size != self.values.size:
            raise ValueError("indices and values must have the same length")
        if self.indices.size > 0:
            if np.any(np.diff(self.indices) <= 0):
                raise TypeError("Indices must be strictly increasing")
            if self.indices[-1] >= self.size:
                raise ValueError("Index out of bounds")

    def numNonzeros(self) -> int:
        # a METHOD, matching pyspark.ml.linalg.SparseVector.numNonzeros()
        return int(np.count_nonzero(self.values))

    def toArray(self) -> np.ndarray:
        arr = np.zeros(self.size, dtype=np.float64)
        arr[self.indices] = self.values
        return arr

    def dot(self, other) -> float:
        if isinstance(other, SparseVector):
            return float(np.dot(self.toArray(), other.toArray()))
        if isinstance(other, DenseVector):
            other = other.values
        other = np.asarray(other, dtype=np.float64)
        return float(np.dot(other[self.indices], self.values))

    def norm(self, p) -> float:
        return float(np.linalg.norm(self.values, p))

    def squared_distance(self, other) -> float:
        return float(np.sum((self.toArray() - _to_array(other)) ** 2))

    def __len__(self) -> int:
        return self.size

    def __getitem__(self, index: int) -> float:
        if index < 0:
            index += self.size
        if index >= self.size or index < 0:
            raise IndexError(f"Index {index} out of bounds.")
        insert = np.searchsorted(self.indices, index)
        if insert < len(self.indices) and self.indices[insert] == index:
            return float(self.values[insert])
        return 0.0

    def __eq__(self, other) -> bool:
        if isinstance(other, SparseVector):
            return (
                self.size == other.size
                and np.array_equal(self.indices, other.indices)
                and np.array_equal(self.values, other.values)
            )
        if isinstance(other, DenseVector):
            return self.size == len(other) and np.array_equal(self.toArray(), other.values)
        return False

    def __hash__(self):
        return DenseVector(self.toArray()).__hash__()

    def __repr__(self) -> str:
        inds = ", ".join(str(int(i)) for i in self.indices)
        vals = ", ".join(_format_float(v) for v in self.values)
        return f"SparseVector({self.size}, {{{', '.join(f'{int(i)}: {_format_float(v)}' for i, v in zip(self.indices, self.values))}}})"

    def __str__(self) -> str:
        inds = ",".join(str(int(i)) for i in self.indices)
        vals = ",".join(_format_float(v) for v in self.values)
        return f"({self.size},[{inds}],[{vals}])"


class Matrix:
    def toArray(self) -> np.ndarray:
        raise NotImplementedError


class DenseMatrix(Matrix):
    """Column-major dense matrix (Spark stores matrices column-major)."""

    def __init__(self, numRows: int, numCols: int, values, isTransposed: bool = False):
        self.numRows = int(numRows)
        self.numCols = int(numCols)
        self.values = np.asarray(values, dtype=np.float64).ravel()
        self.isTransposed = bool(isTransposed)
        if self.values.size != self.numRows * self.numCols:
            raise ValueError("values length must be numRows*numCols")

    def toArray(self) -> np.ndarray:
        if self.isTransposed:
            return self.values.reshape(self.numRows, self.numCols)
        return self.values.reshape(self.numCols, self.numRows).T

    def __getitem__(self, idx):
        i, j = idx
        return self.toArray()[i, j]

    def __eq__(self, other):
        return isinstance(other, Matrix) and np.array_equal(self.toArray(), other.toArray())

    def __repr__(self):
        return f"DenseMatrix({self.numRows}, {self.numCols}, {self.values.tolist()}, {self.isTransposed})"


class SparseMatrix(Matrix):
    """CSC sparse matrix, pyspark.ml.linalg.SparseMatrix layout."""

    def __init__(self, numRows, numCols, colPtrs, rowIndices, values, isTransposed=False):
        self.numRows = int(numRows)
        self.numCols = int(numCols)
        self.colPtrs = np.asarray(colPtrs, dtype=np.int32)
        self.rowIndices = np.asarray(rowIndices, dtype=np.int32)
        self.values = np.asarray(values, dtype=np.float64)
        self.isTransposed = bool(isTransposed)

    def toArray(self) -> np.ndarray:
        arr = np.zeros((self.numRows, self.numCols), dtype=np.float64)
        if self.isTransposed:
            for i in range(self.numRows):
                for k in range(self.colPtrs[i], self.colPtrs[i + 1]):
                    arr[i, self.rowIndices[k]] = self.values[k]
        else:
            for j in range(self.numCols):
                for k in range(self.colPtrs[j], self.colPtrs[j + 1]):
                    arr[self.rowIndices[k], j] = self.values[k]
        return arr

    def __eq__(self, other):
        return isinstance(other, Matrix) and np.array_equal(self.toArray(), other.toArray())


class Vectors:
    """Factory (pyspark.ml.linalg.Vectors).

    >>> Vectors.dense([1.0, 0.0, 3.0])
    DenseVector([1.0, 0.0, 3.0])
    >>> Vectors.sparse(3, [0, 2], [1.0, 3.0]).toArray().tolist()
    [1.0, 0.0, 3.0]
    """

    @staticmethod
    def dense(*elements) -> DenseVector:
        if len(elements) == 1 and not np.isscalar(elements[0]):
            elements = elements[0]
        return DenseVector(np.asarray(elements, dtype=np.float64))

    @staticmethod
    def sparse(size: int, *args) -> SparseVector:
        return SparseVector(size, *args)

    @staticmethod
    def zeros(size: int) -> DenseVector:
        return DenseVector(np.zeros(size))


class Matrices:
    @staticmethod
    def dense(numRows: int, numCols: int, values) -> DenseMatrix:
        return DenseMatrix(numRows, numCols, values)

    @staticmethod
    def sparse(numRows, numCols, colPtrs, rowIndices, values) -> SparseMatrix:
        return SparseMatrix(numRows, numCols, colPtrs, rowIndices, values)


def _to_array(v) -> np.ndarray:
    if isinstance(v, Vector):
        return v.toArray()
    return np.asarray(v, dtype=np.float64)


def _format_float(v: float) -> str:
    s = repr(float(v))
    return s
