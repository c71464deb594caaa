"""RuntimeConfig — tier-2 runtime knobs.

Estimator ``Params`` are tier 1 (the compat contract, serialized). This is
the small non-serialized runtime tier: device selection, HBM budget, dtype
and kernel-vs-torch A/B toggles. Sourced from env vars, overridable in code.
No global mutable config object is exported — construct where needed.
"""

from __future__ import annotations

import os
from dataclasses import dataclass, field


def _env_flag(name: str, default: bool) -> bool:
    v = os.environ.get(name)
    return default if v is None else v not in ("0", "false", "False", "")


@dataclass
class RuntimeConfig:
    device: str = field(default_factory=lambda: os.environ.get("FLOWML_DEVICE", "auto"))
    hbm_budget_fraction: float = field(
        default_factory=lambda: float(os.environ.get("FLOWML_HBM_FRACTION", "0.9"))
    )
    use_bf16: bool = field(default_factory=lambda: _env_flag("FLOWML_BF16", True))
    force_torch_ops: bool = field(
        default_factory=lambda: _env_flag("FLOWML_FORCE_TORCH_OPS", False)
    )
    log_dir: str = field(default_factory=lambda: os.environ.get("FLOWML_LOG_DIR", ""))

    def resolve_device(self) -> str:
        if self.device != "auto":
            return self.device
        import torch

        if torch.cuda.is_available():
            local_rank = int(os.environ.get("LOCAL_RANK", "0"))
            return f"cuda:{local_rank}"
        return "cpu"
