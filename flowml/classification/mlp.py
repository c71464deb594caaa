"""MultilayerPerceptronClassifier / Model.

Behavioral spec: ``spark:.../ml/classification/MultilayerPerceptronClassifier.scala``
+ ``ml/ann/Layer.scala``: ``layers=[in, hidden..., out]``
topology, solver ``l-bfgs`` default (also ``gd``), ``maxIter=100``,
``tol=1e-6``, ``blockSize=128`` (kept for API parity — block-stacking is a
JVM-BLAS artifact; here the whole HBM-resident shard is one batched GEMM),
softmax + cross-entropy output. Divergence, per the north star
(BASELINE.json:5): the hidden activation is **ReLU** (Spark uses sigmoid);
``activation="sigmoid"`` restores Spark's choice.

Model state is the flat weight vector in Spark's layout — per affine layer a
column-major (numOut×numIn) W block then the bias — so persistence
round-trips (SURVEY.md §2.2 persistence row: MLP saves ``{layers[], weights}``).

MI355X mapping: forward/backward are the K5-K7 fused
MFMA GEMM kernels on GPU (bf16 compute / fp32 accumulate); the flat gradient
is one fused RCCL all_reduce (C3) per evaluation; every rank then takes the
identical optimizer step.
"""

from __future__ import annotations

from typing import List, Optional

import torch

from flowml import dist, ops
from flowml.classification.summary import _HasTrainingSummary
from flowml.classification.base import (
    ProbabilisticClassificationModel,
    ProbabilisticClassifier,
    _features_tensor,
    _label_tensor,
)
from flowml.frame import FlowFrame
from flowml.linalg import DenseVector
from flowml.optim import lbfgs
from flowml.param import (
    HasMaxIter,
    HasSeed,
    HasTol,
    Param,
    TypeConverters,
)


class _MLPParams(HasMaxIter, HasTol, HasSeed):
    layers: Param[List[int]] = Param(
        "undefined",
        "layers",
        "Sizes of layers from input layer to output layer.",
        TypeConverters.toListInt,
    )
    blockSize: Param[int] = Param(
        "undefined",
        "blockSize",
        "Block size for stacking input data in matrices (API parity; the GPU "
        "path batches the whole shard).",
        TypeConverters.toInt,
    )
    solver: Param[str] = Param(
        "undefined", "solver", "The solver algorithm for optimization: l-bfgs, gd.", TypeConverters.toString
    )
    stepSize: Param[float] = Param(
        "undefined", "stepSize", "Step size to be used for each iteration of optimization (gd).", TypeConverters.toFloat
    )
    activation: Param[str] = Param(
        "undefined",
        "activation",
        "Hidden activation: relu (flowml default, north-star spec) or sigmoid (Spark parity).",
        TypeConverters.toString,
    )

    def __init__(self):
        super().__init__()
        self._setDefault(
            maxIter=100,
            tol=1e-6,
            blockSize=128,
            solver="l-bfgs",
            stepSize=0.03,
            activation="relu",
            seed=17,
        )

    def getLayers(self) -> List[int]:
        return self.getOrDefault(self.layers)

    def getSolver(self) -> str:
        return self.getOrDefault(self.solver)

    def getActivation(self) -> str:
        return self.getOrDefault(self.activation)


def _unflatten(flat: torch.Tensor, layers: List[int]):
    """Spark ANN layout: per layer, column-major (out×in) W then bias."""
    Ws, bs = [], []
    off = 0
    for i in range(len(layers) - 1):
        fin, fout = layers[i], layers[i + 1]
        W = flat[off : off + fin * fout].view(fin, fout).T.contiguous()  # col-major (out,in)
        off += fin * fout
        b = flat[off : off + fout].contiguous()
        off += fout
        Ws.append(W)
        bs.append(b)
    return Ws, bs


def _flatten(Ws, bs) -> torch.Tensor:
    parts = []
    for W, b in zip(Ws, bs):
        parts.append(W.T.reshape(-1))  # column-major
        parts.append(b.reshape(-1))
    return torch.cat(parts)


def n_weights(layers: List[int]) -> int:
    return sum(layers[i] * layers[i + 1] + layers[i + 1] for i in range(len(layers) - 1))


class MultilayerPerceptronClassifier(_MLPParams, ProbabilisticClassifier):
    def __init__(
        self,
        *,
        featuresCol: Optional[str] = None,
        labelCol: Optional[str] = None,
        predictionCol: Optional[str] = None,
        rawPredictionCol: Optional[str] = None,
        probabilityCol: Optional[str] = None,
        layers: Optional[List[int]] = None,
        blockSize: Optional[int] = None,
        solver: Optional[str] = None,
        maxIter: Optional[int] = None,
        tol: Optional[float] = None,
        seed: Optional[int] = None,
        stepSize: Optional[float] = None,
        activation: Optional[str] = None,
    ):
        super().__init__()
        self._set(
            featuresCol=featuresCol,
            labelCol=labelCol,
            predictionCol=predictionCol,
            rawPredictionCol=rawPredictionCol,
            probabilityCol=probabilityCol,
            layers=layers,
            blockSize=blockSize,
            solver=solver,
            maxIter=maxIter,
            tol=tol,
            seed=seed,
            stepSize=stepSize,
            activation=activation,
        )

    def setLayers(self, value: List[int]):
        return self.set(self.layers, value)

    def _fit(self, dataset: FlowFrame) -> "MultilayerPerceptronClassificationModel":
        layers = self.getLayers()
        if not layers or len(layers) < 2:
            raise ValueError("layers must have at least input and output sizes")
        X = _features_tensor(dataset, self.getFeaturesCol()).to(torch.float32)
        y = _label_tensor(dataset, self.getLabelCol(), classification=True).to(torch.int64)
        dev = X.device
        if X.shape[1] != layers[0]:
            raise ValueError(
                f"features dim {X.shape[1]} != layers[0] {layers[0]}"
            )
        activation = self.getActivation()
        # north-star compute dtype on GPU is bf16 (BASELINE.json:8): cast and
        # pre-pad the shard once; master weights/optimizer state stay fp32
        use_bf16 = X.is_cuda
        if use_bf16:
            K0 = layers[0]
            Kp = ((K0 + 31) // 32) * 32
            Xc = torch.zeros(X.shape[0], Kp, dtype=torch.bfloat16, device=dev)
            Xc[:, :K0] = X.to(torch.bfloat16)
        else:
            Xc = X

        # global row count (loss normalizer)
        cnt = torch.tensor([float(len(X))], dtype=torch.float64, device=dev)
        dist.all_reduce_(cnt)
        total = max(float(cnt.item()), 1.0)

        gen = torch.Generator(device="cpu").manual_seed(self.getOrDefault(self.seed))
        flat0_parts = []
        for i in range(len(layers) - 1):
            fin, fout = layers[i], layers[i + 1]
            # He-style init for relu, Glorot for sigmoid (host RNG → identical
            # replicas on every rank without a broadcast)
            if activation == "relu":
                scale = (2.0 / fin) ** 0.5
            else:
                scale = (6.0 / (fin + fout)) ** 0.5
            Wl = torch.randn(fin * fout, generator=gen) * scale
            flat0_parts.append(Wl)
            flat0_parts.append(torch.zeros(fout))
        theta0 = torch.cat(flat0_parts).to(dev, torch.float32)

        def f_grad(theta: torch.Tensor):
            Ws, bs = _unflatten(theta, layers)
            if use_bf16:
                Ws = [w.to(torch.bfloat16) for w in Ws]
            loss_s, gWs, gBs = ops.mlp_grad(
                Xc, y, Ws, bs, activation=activation, in_features=layers[0]
            )
            flat_g = _flatten([g.to(torch.float32) for g in gWs], gBs)
            buf = torch.cat([loss_s.reshape(1), flat_g])
            dist.all_reduce_(buf)
            return float(buf[0]) / total, buf[1:] / total

        solver = self.getSolver()
        max_iter = self.getMaxIter()
        if solver == "l-bfgs":
            theta, history = lbfgs(f_grad, theta0, max_iter=max_iter, tol=self.getTol())
        elif solver == "gd":
            lr = self.getOrDefault(self.stepSize)
            theta = theta0.clone()
            history = []
            for _ in range(max_iter):
                loss, g = f_grad(theta)
                history.append(loss)
                theta -= lr * g
        else:
            raise ValueError(f"Unsupported solver {solver!r}")

        model = MultilayerPerceptronClassificationModel(
            layers=list(layers), weights=theta.detach().cpu(), activation=activation
        )
        self._copyValues(model)
        model._set_parent(self)
        from flowml.classification.summary import ClassificationTrainingSummary

        model._summary = ClassificationTrainingSummary(
            model, dataset, self.getLabelCol(), self.getPredictionCol()
        )
        model._objective_history = [float(h) for h in history]
        return model


class MultilayerPerceptronClassificationModel(_MLPParams, _HasTrainingSummary, ProbabilisticClassificationModel):
    def __init__(
        self,
        layers: Optional[List[int]] = None,
        weights: Optional[torch.Tensor] = None,
        activation: str = "relu",
    ):
        super().__init__()
        self._layers = list(layers) if layers else []
        self._weights = weights if weights is not None else torch.zeros(0)
        if layers:
            self._set(layers=list(layers))
        self._set(activation=activation)
        self.numClasses = self._layers[-1] if self._layers else 0
        self.numFeatures = self._layers[0] if self._layers else 0
        self._objective_history: List[float] = []
        self._cached_device = None
        self._cached = None

    @property
    def weights(self) -> DenseVector:
        return DenseVector(self._weights.to(torch.float64).cpu().numpy())

    def _mats(self, device, dtype=torch.float32):
        if self._cached is not None and self._cached_device == (device, dtype):
            return self._cached
        flat = self._weights.to(device=device, dtype=dtype)
        self._cached = _unflatten(flat, self._layers)
        self._cached_device = (device, dtype)
        return self._cached

    def predictRaw(self, features: torch.Tensor) -> torch.Tensor:
        Ws, bs = self._mats(features.device)
        return ops.mlp_forward(features.to(torch.float32), Ws, bs, activation=self.getActivation())

    def raw2probability(self, raw: torch.Tensor) -> torch.Tensor:
        return torch.softmax(raw, dim=1)

    def write(self):
        from flowml.io import GenericWriter

        return GenericWriter(self)

    def save(self, path: str):
        self.write().save(path)

    @classmethod
    def load(cls, path: str):
        from flowml.io import load as _load

        return _load(path)
