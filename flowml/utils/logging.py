"""Structured per-rank JSONL metrics logging.

Replaces the reference stack's log4j/Codahale/Spark-UI observability with
``logs/rank{k}.jsonl`` records: per-iteration loss, phase timings, memory.
"""

from __future__ import annotations

import json
import os
import time
from typing import Any, Dict, Optional


class MetricsLogger:
    def __init__(self, log_dir: Optional[str] = None, rank: int = 0, enabled: bool = True):
        self.rank = rank
        self.enabled = enabled and log_dir is not None
        self._f = None
        if self.enabled:
            os.makedirs(log_dir, exist_ok=True)
            self._f = open(os.path.join(log_dir, f"rank{rank}.jsonl"), "a")

    def log(self, event: str, **fields: Any) -> None:
        if not self.enabled:
            return
        rec: Dict[str, Any] = {"ts": time.time(), "rank": self.rank, "event": event}
        rec.update(fields)
        self._f.write(json.dumps(rec) + "\n")
        self._f.flush()

    def close(self) -> None:
        if self._f:
            self._f.close()
            self._f = None


_global: Optional[MetricsLogger] = None


def get_logger() -> MetricsLogger:
    global _global
    if _global is None:
        log_dir = os.environ.get("FLOWML_LOG_DIR")
        rank = int(os.environ.get("RANK", "0"))
        _global = MetricsLogger(log_dir, rank=rank, enabled=log_dir is not None)
    return _global
