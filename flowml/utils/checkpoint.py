"""Training checkpoint/resume.

Two mechanisms mirror the reference stack:
1. model persistence — the user-facing Spark save format (``flowml.io``);
2. periodic *training* checkpoints inside long boosting runs — the analog of
   Spark's ``PeriodicRDDCheckpointer`` inside GBT (every ``checkpointInterval``
   rounds): rank 0 writes {round, ensemble-so-far in the save format}, and a
   run can resume with ``GBTClassifier.setInitModel(load_gbt_checkpoint(...))``.
"""

from __future__ import annotations

import json
import os
import shutil
from typing import List, Optional


def save_gbt_checkpoint(ckpt_dir: str, trees, weights: List[float], rounds: int, num_features: int) -> str:
    """Write checkpoint atomically (tmp dir + rename); keeps only the latest."""
    from flowml.classification.gbt import GBTClassificationModel

    os.makedirs(ckpt_dir, exist_ok=True)
    tmp = os.path.join(ckpt_dir, f".tmp_round_{rounds}")
    final = os.path.join(ckpt_dir, f"round_{rounds:06d}")
    if os.path.exists(tmp):
        shutil.rmtree(tmp)
    model = GBTClassificationModel(trees=list(trees), treeWeights=list(weights), numFeatures=num_features)
    model.write().save(tmp)
    with open(os.path.join(tmp, "CHECKPOINT.json"), "w") as f:
        json.dump({"round": rounds, "num_trees": len(trees)}, f)
    if os.path.exists(final):
        shutil.rmtree(final)
    os.rename(tmp, final)
    # prune older checkpoints (keep latest 2, like PeriodicCheckpointer)
    kept = sorted(d for d in os.listdir(ckpt_dir) if d.startswith("round_"))
    for d in kept[:-2]:
        shutil.rmtree(os.path.join(ckpt_dir, d), ignore_errors=True)
    return final


def latest_gbt_checkpoint(ckpt_dir: str) -> Optional[str]:
    if not os.path.isdir(ckpt_dir):
        return None
    rounds = sorted(d for d in os.listdir(ckpt_dir) if d.startswith("round_"))
    return os.path.join(ckpt_dir, rounds[-1]) if rounds else None


def load_gbt_checkpoint(path: str):
    """Load a checkpoint dir → (GBTClassificationModel, round)."""
    from flowml.io import load

    model = load(path)
    with open(os.path.join(path, "CHECKPOINT.json")) as f:
        info = json.load(f)
    return model, int(info["round"])
