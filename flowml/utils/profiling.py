"""Phase timers + chrome-trace dump.

Replaces the Spark UI/event-log with hipEvent-pair phase timing: every
``with timer.phase("hist"):`` records a GPU-accurate duration (torch.cuda
events on ROCm; wall time on CPU) and ``dump_chrome_trace`` writes a
chrome://tracing-loadable JSON per rank. rocprofv3 counter runs wrap ``bench.py``
directly (see profiles/NOTES.md for the recipes).
"""

from __future__ import annotations

import json
import time
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch


class PhaseTimer:
    def __init__(self, enabled: bool = True, use_cuda: Optional[bool] = None):
        self.enabled = enabled
        self.use_cuda = torch.cuda.is_available() if use_cuda is None else use_cuda
        self.events: List[Dict] = []
        self._pending: List = []

    @contextmanager
    def phase(self, name: str):
        if not self.enabled:
            yield
            return
        if self.use_cuda:
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            t0_wall = time.perf_counter()
            e0.record()
            try:
                yield
            finally:
                e1.record()
                self._pending.append((name, t0_wall, e0, e1))
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                t1 = time.perf_counter()
                self.events.append(
                    {"name": name, "ts": t0 * 1e6, "dur": (t1 - t0) * 1e6}
                )

    def flush(self) -> None:
        """Resolve pending CUDA event pairs (synchronizes)."""
        if self._pending:
            torch.cuda.synchronize()
            for name, t0_wall, e0, e1 in self._pending:
                self.events.append(
                    {"name": name, "ts": t0_wall * 1e6, "dur": e0.elapsed_time(e1) * 1e3}
                )
            self._pending.clear()

    def totals(self) -> Dict[str, float]:
        self.flush()
        out: Dict[str, float] = {}
        for e in self.events:
            out[e["name"]] = out.get(e["name"], 0.0) + e["dur"] / 1e3
        return out  # ms per phase

    def dump_chrome_trace(self, path: str, rank: int = 0) -> None:
        self.flush()
        trace = [
            {
                "name": e["name"],
                "cat": "flowml",
                "ph": "X",
                "pid": rank,
                "tid": 0,
                "ts": e["ts"],
                "dur": e["dur"],
            }
            for e in self.events
        ]
        with open(path, "w") as f:
            json.dump({"traceEvents": trace}, f)
