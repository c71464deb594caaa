"""Params system — typed estimator/transformer parameters.

Behavioral spec: the ``pyspark.ml.param`` package (upstream
``spark:python/pyspark/ml/param/__init__.py``).
The reference mount was empty, so parity is to the public
pyspark.ml API: ``Param`` objects hang off a ``Params`` owner, user-set values
(``_paramMap``) are kept distinct from defaults (``_defaultParamMap``) because
the save format serializes them separately (SURVEY.md §2.2 persistence row),
and ``uid`` generation follows Spark's ``<ClassName>_<hex>`` convention.
"""

from __future__ import annotations

import copy as _copy
import uuid
from typing import Any, Callable, Dict, Generic, List, Optional, TypeVar

T = TypeVar("T")


class Param(Generic[T]):
    """A named parameter with documentation and an optional type converter.

    Identity semantics match pyspark: a Param belongs to a ``Params`` instance
    (its ``parent`` is the owner's uid) and equality is (parent, name).
    """

    def __init__(
        self,
        parent: "Params",
        name: str,
        doc: str,
        typeConverter: Optional[Callable[[Any], T]] = None,
    ):
        self.parent = parent.uid if isinstance(parent, Params) else str(parent)
        self.name = str(name)
        self.doc = str(doc)
        self.typeConverter = typeConverter

    def _copy_new_parent(self, parent: "Params") -> "Param[T]":
        p = _copy.copy(self)
        p.parent = parent.uid
        return p

    def __str__(self) -> str:
        return f"{self.parent}__{self.name}"

    def __repr__(self) -> str:
        return f"Param(parent={self.parent!r}, name={self.name!r}, doc={self.doc!r})"

    def __hash__(self) -> int:
        return hash(str(self))

    def __eq__(self, other: Any) -> bool:
        if isinstance(other, Param):
            return self.parent == other.parent and self.name == other.name
        return False


# ---------------------------------------------------------------------------
# Type converters (subset of pyspark.ml.param.TypeConverters)
# ---------------------------------------------------------------------------


class TypeConverters:
    @staticmethod
    def toInt(value: Any) -> int:
        if isinstance(value, bool):
            raise TypeError(f"Could not convert {value} to int")
        iv = int(value)
        if float(iv) != float(value):
            raise TypeError(f"Could not convert {value} to int")
        return iv

    @staticmethod
    def toFloat(value: Any) -> float:
        return float(value)

    @staticmethod
    def toBoolean(value: Any) -> bool:
        if isinstance(value, bool):
            return value
        raise TypeError(f"Boolean Param requires value of type bool. Found {type(value)}.")

    @staticmethod
    def toString(value: Any) -> str:
        if isinstance(value, str):
            return value
        raise TypeError(f"Could not convert {value} to string type")

    @staticmethod
    def toList(value: Any) -> list:
        if isinstance(value, (list, tuple)):
            return list(value)
        raise TypeError(f"Could not convert {value} to list")

    @staticmethod
    def toListInt(value: Any) -> List[int]:
        return [TypeConverters.toInt(v) for v in TypeConverters.toList(value)]

    @staticmethod
    def toListFloat(value: Any) -> List[float]:
        return [float(v) for v in TypeConverters.toList(value)]

    @staticmethod
    def toListString(value: Any) -> List[str]:
        return [TypeConverters.toString(v) for v in TypeConverters.toList(value)]

    @staticmethod
    def toVector(value: Any):
        from flowml import linalg

        if isinstance(value, linalg.Vector):
            return value
        if isinstance(value, (list, tuple)):
            return linalg.Vectors.dense(list(value))
        raise TypeError(f"Could not convert {value} to vector")


class Params:
    """Base class for components that carry Params.

    Mirrors ``pyspark.ml.param.Params``: per-instance ``_paramMap`` (user-set)
    and ``_defaultParamMap`` (class defaults) with getOrDefault resolution
    order user-set → default → KeyError.
    """

    def __init__(self) -> None:
        self.uid = self._randomUID()
        self._paramMap: Dict[Param, Any] = {}
        self._defaultParamMap: Dict[Param, Any] = {}
        # re-bind class-level Param declarations to this instance
        for name in dir(type(self)):
            attr = getattr(type(self), name, None)
            if isinstance(attr, Param):
                setattr(self, name, attr._copy_new_parent(self))

    @classmethod
    def _randomUID(cls) -> str:
        return f"{cls.__name__}_{uuid.uuid4().hex[:16]}"

    # -- param discovery ----------------------------------------------------

    @property
    def params(self) -> List[Param]:
        """All Params of this instance, sorted by name."""
        # instance __dict__ only: __init__ re-bound every class-level Param
        # into the instance, and this avoids triggering properties via getattr
        return sorted(
            (v for v in self.__dict__.values() if isinstance(v, Param)),
            key=lambda p: p.name,
        )

    def __repr__(self) -> str:
        # pyspark shows the uid (e.g. "LogisticRegression_4db9da9f2932")
        return self.uid

    def hasParam(self, paramName: str) -> bool:
        p = getattr(self, paramName, None)
        return isinstance(p, Param)

    def getParam(self, paramName: str) -> Param:
        p = getattr(self, paramName, None)
        if not isinstance(p, Param):
            raise ValueError(f"Cannot find param with name: {paramName}")
        return p

    def _resolveParam(self, param) -> Param:
        if isinstance(param, Param):
            self._shouldOwn(param)
            return param
        if isinstance(param, str):
            return self.getParam(param)
        raise TypeError(f"Cannot resolve {param!r} as a param")

    def _shouldOwn(self, param: Param) -> None:
        if not (param.parent == self.uid and self.hasParam(param.name)):
            raise ValueError(f"Param {param} does not belong to {self.uid}.")

    # -- get/set ------------------------------------------------------------

    def isSet(self, param) -> bool:
        return self._resolveParam(param) in self._paramMap

    def hasDefault(self, param) -> bool:
        return self._resolveParam(param) in self._defaultParamMap

    def isDefined(self, param) -> bool:
        return self.isSet(param) or self.hasDefault(param)

    def getOrDefault(self, param):
        param = self._resolveParam(param)
        if param in self._paramMap:
            return self._paramMap[param]
        if param in self._defaultParamMap:
            return self._defaultParamMap[param]
        raise KeyError(f"Failed to find a default value for {param.name}")

    def set(self, param: Param, value: Any) -> "Params":
        param = self._resolveParam(param)
        if param.typeConverter is not None:
            value = param.typeConverter(value)
        self._paramMap[param] = value
        return self

    def _set(self, **kwargs: Any) -> "Params":
        """Set user-supplied params (skips None, matching pyspark keyword args)."""
        for name, value in kwargs.items():
            if value is None:
                continue
            p = self.getParam(name)
            if p.typeConverter is not None:
                value = p.typeConverter(value)
            self._paramMap[p] = value
        return self

    def _setDefault(self, **kwargs: Any) -> "Params":
        for name, value in kwargs.items():
            p = self.getParam(name)
            if value is not None and p.typeConverter is not None:
                value = p.typeConverter(value)
            self._defaultParamMap[p] = value
        return self

    def clear(self, param: Param) -> "Params":
        param = self._resolveParam(param)
        self._paramMap.pop(param, None)
        return self

    # -- inspection ---------------------------------------------------------

    def explainParam(self, param) -> str:
        param = self._resolveParam(param)
        values = []
        if self.isSet(param):
            values.append(f"current: {self.getOrDefault(param)}")
        if param in self._defaultParamMap:
            values.append(f"default: {self._defaultParamMap[param]}")
        valueStr = f" ({', '.join(values)})" if values else " (undefined)"
        return f"{param.name}: {param.doc}{valueStr}"

    def explainParams(self) -> str:
        return "\n".join(self.explainParam(p) for p in self.params)

    def extractParamMap(self, extra: Optional[Dict[Param, Any]] = None) -> Dict[Param, Any]:
        paramMap = dict(self._defaultParamMap)
        paramMap.update(self._paramMap)
        if extra:
            paramMap.update(extra)
        return paramMap

    # -- copy ---------------------------------------------------------------

    def copy(self, extra: Optional[Dict[Param, Any]] = None) -> "Params":
        """Shallow copy with the same uid and param maps, plus ``extra`` overrides."""
        that = _copy.copy(self)
        that._paramMap = dict(self._paramMap)
        that._defaultParamMap = dict(self._defaultParamMap)
        if extra:
            for p, v in extra.items():
                that._paramMap[that._resolveParam(p.name if isinstance(p, Param) else p)] = v
        return that

    def _copyValues(self, to: "Params", extra: Optional[Dict[Param, Any]] = None) -> "Params":
        """Copy param values from this instance to ``to`` (used fit -> Model)."""
        for p, v in self._defaultParamMap.items():
            if to.hasParam(p.name):
                to._defaultParamMap[to.getParam(p.name)] = v
        paramMap = dict(self._paramMap)
        if extra:
            paramMap.update({self._resolveParam(p): v for p, v in extra.items()})
        for p, v in paramMap.items():
            if to.hasParam(p.name):
                to._paramMap[to.getParam(p.name)] = v
        return to


# ---------------------------------------------------------------------------
# Shared param mixins (pyspark.ml.param.shared equivalents)
# ---------------------------------------------------------------------------


class HasInputCol(Params):
    inputCol: Param[str] = Param(
        "undefined", "inputCol", "input column name.", TypeConverters.toString
    )

    def getInputCol(self) -> str:
        return self.getOrDefault(self.inputCol)

    def setInputCol(self, value: str):
        return self.set(self.inputCol, value)


class HasInputCols(Params):
    inputCols: Param[List[str]] = Param(
        "undefined", "inputCols", "input column names.", TypeConverters.toListString
    )

    def getInputCols(self) -> List[str]:
        return self.getOrDefault(self.inputCols)

    def setInputCols(self, value: List[str]):
        return self.set(self.inputCols, value)


class HasOutputCol(Params):
    outputCol: Param[str] = Param(
        "undefined", "outputCol", "output column name.", TypeConverters.toString
    )

    def getOutputCol(self) -> str:
        return self.getOrDefault(self.outputCol)

    def setOutputCol(self, value: str):
        return self.set(self.outputCol, value)


class HasOutputCols(Params):
    outputCols: Param[List[str]] = Param(
        "undefined", "outputCols", "output column names.", TypeConverters.toListString
    )

    def getOutputCols(self) -> List[str]:
        return self.getOrDefault(self.outputCols)

    def setOutputCols(self, value: List[str]):
        return self.set(self.outputCols, value)


class HasLabelCol(Params):
    labelCol: Param[str] = Param(
        "undefined", "labelCol", "label column name.", TypeConverters.toString
    )

    def getLabelCol(self) -> str:
        return self.getOrDefault(self.labelCol)

    def setLabelCol(self, value: str):
        return self.set(self.labelCol, value)


class HasFeaturesCol(Params):
    featuresCol: Param[str] = Param(
        "undefined", "featuresCol", "features column name.", TypeConverters.toString
    )

    def getFeaturesCol(self) -> str:
        return self.getOrDefault(self.featuresCol)

    def setFeaturesCol(self, value: str):
        return self.set(self.featuresCol, value)


class HasPredictionCol(Params):
    predictionCol: Param[str] = Param(
        "undefined", "predictionCol", "prediction column name.", TypeConverters.toString
    )

    def getPredictionCol(self) -> str:
        return self.getOrDefault(self.predictionCol)

    def setPredictionCol(self, value: str):
        return self.set(self.predictionCol, value)


class HasRawPredictionCol(Params):
    rawPredictionCol: Param[str] = Param(
        "undefined",
        "rawPredictionCol",
        "raw prediction (a.k.a. confidence) column name.",
        TypeConverters.toString,
    )

    def getRawPredictionCol(self) -> str:
        return self.getOrDefault(self.rawPredictionCol)

    def setRawPredictionCol(self, value: str):
        return self.set(self.rawPredictionCol, value)


class HasThresholds(Params):
    thresholds: Param[list] = Param(
        "undefined",
        "thresholds",
        "Thresholds in multi-class classification to adjust the probability "
        "of predicting each class. Array must have length equal to the "
        "number of classes, with values > 0 excepting that at most one "
        "value may be 0. The class with largest value p/t is predicted, "
        "where p is the original probability of that class and t is the "
        "class's threshold.",
        TypeConverters.toListFloat,
    )

    def getThresholds(self) -> list:
        return self.getOrDefault(self.thresholds)

    def setThresholds(self, value):
        return self.set(self.thresholds, value)


class HasProbabilityCol(Params):
    probabilityCol: Param[str] = Param(
        "undefined",
        "probabilityCol",
        "Column name for predicted class conditional probabilities.",
        TypeConverters.toString,
    )

    def getProbabilityCol(self) -> str:
        return self.getOrDefault(self.probabilityCol)

    def setProbabilityCol(self, value: str):
        return self.set(self.probabilityCol, value)


class HasMaxIter(Params):
    maxIter: Param[int] = Param(
        "undefined", "maxIter", "max number of iterations (>= 0).", TypeConverters.toInt
    )

    def getMaxIter(self) -> int:
        return self.getOrDefault(self.maxIter)

    def setMaxIter(self, value: int):
        return self.set(self.maxIter, value)


class HasTol(Params):
    tol: Param[float] = Param(
        "undefined",
        "tol",
        "the convergence tolerance for iterative algorithms (>= 0).",
        TypeConverters.toFloat,
    )

    def getTol(self) -> float:
        return self.getOrDefault(self.tol)

    def setTol(self, value: float):
        return self.set(self.tol, value)


class HasSeed(Params):
    seed: Param[int] = Param("undefined", "seed", "random seed.", TypeConverters.toInt)

    def getSeed(self) -> int:
        return self.getOrDefault(self.seed)

    def setSeed(self, value: int):
        return self.set(self.seed, value)


class HasWeightCol(Params):
    weightCol: Param[str] = Param(
        "undefined",
        "weightCol",
        "weight column name. If this is not set or empty, we treat all instance weights as 1.0.",
        TypeConverters.toString,
    )

    def getWeightCol(self) -> str:
        return self.getOrDefault(self.weightCol)

    def setWeightCol(self, value: str):
        return self.set(self.weightCol, value)


class HasHandleInvalid(Params):
    handleInvalid: Param[str] = Param(
        "undefined",
        "handleInvalid",
        "how to handle invalid entries. Options are 'skip', 'error' and 'keep'.",
        TypeConverters.toString,
    )

    def getHandleInvalid(self) -> str:
        return self.getOrDefault(self.handleInvalid)

    def setHandleInvalid(self, value: str):
        return self.set(self.handleInvalid, value)
