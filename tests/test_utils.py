"""Aux subsystems: profiling, metrics logging, runtime config."""

import json
import os

import torch


class TestPhaseTimer:
    def test_phases_and_chrome_trace(self, tmp_path):
        from flowml.utils.profiling import PhaseTimer

        t = PhaseTimer(use_cuda=False)
        with t.phase("a"):
            sum(range(1000))
        with t.phase("b"):
            pass
        with t.phase("a"):
            pass
        totals = t.totals()
        assert set(totals) == {"a", "b"}
        assert totals["a"] >= 0 and totals["b"] >= 0
        p = tmp_path / "trace.json"
        t.dump_chrome_trace(str(p), rank=3)
        trace = json.loads(p.read_text())
        assert len(trace["traceEvents"]) == 3
        assert all(e["pid"] == 3 and e["ph"] == "X" for e in trace["traceEvents"])

    def test_disabled_is_noop(self):
        from flowml.utils.profiling import PhaseTimer

        t = PhaseTimer(enabled=False, use_cuda=False)
        with t.phase("x"):
            pass
        assert t.totals() == {}


class TestMetricsLogger:
    def test_jsonl_written_during_fit(self, tmp_path, monkeypatch):
        import numpy as np

        import flowml.utils.logging as fl
        from flowml.classification import LogisticRegression
        from flowml.frame import FlowFrame

        monkeypatch.setenv("FLOWML_LOG_DIR", str(tmp_path))
        fl._global = None  # reset cached logger so the env var is re-read
        r = np.random.default_rng(0)
        X = torch.from_numpy(r.normal(size=(200, 3)).astype(np.float32))
        y = (X[:, 0] > 0).to(torch.float64)
        LogisticRegression(maxIter=5).fit(FlowFrame({"features": X, "label": y}))
        fl.get_logger().close()
        fl._global = None
        log_path = tmp_path / "rank0.jsonl"
        assert log_path.exists()
        rows = [json.loads(l) for l in log_path.read_text().splitlines()]
        assert any(r.get("event") == "lr_fit" for r in rows)


class TestRuntimeConfig:
    def test_env_overrides(self, monkeypatch):
        from flowml.utils.config import RuntimeConfig

        monkeypatch.setenv("FLOWML_DEVICE", "cpu")
        c = RuntimeConfig()
        assert c.resolve_device() == "cpu"
