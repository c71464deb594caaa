"""Streaming micro-batch inference with hipGraph-captured batches (config 5).

Replaces the reference's lazy ``PipelineModel.transform`` + evaluator job
 with an eager micro-batch path: pinned host staging → H2D →
ensemble forward (K8 / K5) → argmax → D2H, with the entire per-batch GPU
sequence captured once into a hipGraph (``torch.cuda.CUDAGraph`` is hipGraph
on ROCm) and replayed per batch — one graph launch instead of per-op launches
(BASELINE.json:11 "hipGraph-captured batches"). Multi-GPU serving
round-robins micro-batches across ranks (each rank runs its own replayer).
"""

from __future__ import annotations

from typing import Optional

import torch

from flowml.classification.base import ClassificationModel


class StreamingClassifier:
    """hipGraph-replayed fixed-shape micro-batch scorer for a fitted model.

    ``infer(X)`` returns (prediction, rawPrediction) for a batch of exactly
    ``batch_size`` rows (pad the tail batch; the pad rows' outputs are valid
    predictions of the padded values and can be sliced off by the caller).
    """

    def __init__(
        self,
        model: ClassificationModel,
        batch_size: int,
        num_features: Optional[int] = None,
        device=None,
        use_graph: bool = True,
        featurizer=None,
        input_col: str = "features",
        feature_col: str = "features",
    ):
        """``featurizer``: optional fitted transformer/PipelineModel of
        NUMERIC, sync-free stages (scalers, Normalizer, Binarizer,
        VectorAssembler over tensor cols...) applied to the raw micro-batch
        INSIDE the captured graph — the SURVEY.md §2.4 K8
        "featurize + ensemble-forward" capture. ``num_features`` then sizes
        the RAW input (``input_col``); ``feature_col`` names the featurized
        output the model consumes. Stages that touch the host (StringIndexer
        on string columns, handleInvalid='error' scans) cannot be captured —
        capture failures fall back to eager replay with a warning.
        """
        self.model = model
        self.batch_size = int(batch_size)
        self.device = torch.device(device) if device is not None else torch.device("cuda:0")
        nf = num_features or getattr(model, "numFeatures", 0)
        if not nf:
            raise ValueError("num_features required (model has no numFeatures)")
        self.num_features = int(nf)
        self.use_graph = bool(use_graph) and self.device.type == "cuda"
        self._featurizer = featurizer
        self._input_col = input_col
        self._feature_col = feature_col

        self._static_in = torch.zeros(
            self.batch_size, self.num_features, dtype=torch.float32, device=self.device
        )
        self._graph = None
        self._static_raw = None
        self._static_pred = None
        if self.use_graph:
            try:
                self._capture()
            except Exception as e:  # noqa: BLE001 - sync-in-capture etc.
                import warnings

                warnings.warn(
                    f"hipGraph capture failed ({type(e).__name__}: {e}); "
                    "falling back to eager micro-batches"
                )
                self._graph = None

    def _forward(self):
        feats = self._static_in
        if self._featurizer is not None:
            from flowml.frame import FlowFrame

            df = FlowFrame({self._input_col: self._static_in},
                           device=self.device)
            feats = self._featurizer.transform(df).col(self._feature_col)
        raw = self.model.predictRaw(feats)
        pred = raw.argmax(dim=1)
        return raw, pred

    def _capture(self) -> None:
        # warm up allocations/kernels outside capture, on a side stream
        s = torch.cuda.Stream(device=self.device)
        s.wait_stream(torch.cuda.current_stream(self.device))
        with torch.cuda.stream(s):
            for _ in range(3):
                self._forward()
        torch.cuda.current_stream(self.device).wait_stream(s)
        torch.cuda.synchronize(self.device)
        self._graph = torch.cuda.CUDAGraph()
        # thread_local: lets the RCCL watchdog (if a process group is
        # live) query its work events while we capture
        with torch.cuda.graph(self._graph, capture_error_mode="thread_local"):
            self._static_raw, self._static_pred = self._forward()

    @torch.no_grad()
    def infer(self, X: torch.Tensor):
        """Score one micro-batch (batch_size × num_features)."""
        if X.shape != self._static_in.shape:
            raise ValueError(
                f"batch shape {tuple(X.shape)} != captured {tuple(self._static_in.shape)}"
            )
        self._static_in.copy_(X, non_blocking=True)
        if self._graph is not None:
            self._graph.replay()
            return self._static_pred, self._static_raw
        raw, pred = self._forward()
        return pred, raw

    def infer_stream(self, batches):
        """Generator over a stream of host/device batches (pipelined H2D)."""
        for X in batches:
            pred, raw = self.infer(X.to(self.device, non_blocking=True))
            yield pred


class StreamingDriver:
    """Multi-rank streaming orchestrator (BASELINE config 5, 8-GPU path).

    Round-robins a global micro-batch stream across ranks — rank r scores
    batches i with ``i % world == r`` on its own :class:`StreamingClassifier`
    (hipGraph replays stay rank-local; nothing crosses GPUs per batch) — and
    aggregates at the end: processed-row and per-class prediction counts are
    summed with one ``all_reduce`` so the caller gets GLOBAL stream totals.
    VERDICT r1 next-round #6: the measured aggregate replaces the r1
    single-GPU extrapolation.
    """

    def __init__(self, classifier: "StreamingClassifier"):
        from flowml import dist

        self.sc = classifier
        self.rank = dist.get_rank()
        self.world = dist.get_world_size()

    def run(self, batches, n_classes: Optional[int] = None,
            count_classes: bool = True) -> dict:
        """Consume a global stream (iterable of tensors); returns aggregate
        {"rows": global rows scored, "class_counts": global per-class counts,
        "rows_local": this rank's share, "elapsed_s", "flows_per_s"}.

        ``count_classes=False`` skips the per-batch bincount (measured
        ~0.25 ms/262k batch — 30% of the graph-replay time) and returns
        ``class_counts=None``; row counts are still all-reduced.
        """
        import time as _time

        from flowml import dist

        dev = self.sc.device
        C = int(n_classes or getattr(self.sc.model, "numClasses", 2))
        counts = torch.zeros(C, dtype=torch.float64,
                             device=dev if dev.type == "cuda" else "cpu")
        n_local = 0
        dist.barrier()
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)
        t0 = _time.perf_counter()
        for i, X in enumerate(batches):
            if i % self.world != self.rank:
                continue
            pred, _raw = self.sc.infer(X.to(dev, non_blocking=True))
            n_local += pred.numel()
            if count_classes:
                counts += torch.bincount(
                    pred.reshape(-1).to(torch.int64), minlength=C
                ).to(counts.dtype)[:C]
        if dev.type == "cuda":
            torch.cuda.synchronize(dev)
        dist.barrier()
        elapsed = _time.perf_counter() - t0
        agg = torch.cat([counts, counts.new_tensor([float(n_local)])])
        dist.all_reduce_(agg)
        rows = int(agg[-1].item())
        return {
            "rows": rows,
            "rows_local": n_local,
            "class_counts": agg[:-1].cpu().tolist() if count_classes else None,
            "elapsed_s": elapsed,
            "flows_per_s": rows / elapsed if elapsed > 0 else 0.0,
        }
