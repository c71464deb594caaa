"""Failure detection / checkpoint-restart: kill a training
process mid-run, restart from the last periodic checkpoint, and verify the
restarted run reproduces the uninterrupted model bitwise (determinism makes
this checkable)."""

import os
import signal
import subprocess
import sys
import time

import numpy as np
import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

_TRAIN_SCRIPT = """
import sys, numpy as np, torch
sys.path.insert(0, {repo!r})
from flowml import FlowFrame
from flowml.classification import GBTClassifier

r = np.random.default_rng(4)
X = r.normal(size=(3000, 6)).astype(np.float32)
y = ((X[:, 0] * X[:, 1] + X[:, 2]) > 0).astype(np.float64)
df = FlowFrame({{"features": torch.from_numpy(X), "label": y}})
gbt = GBTClassifier(maxIter=16, maxDepth=3, seed=7, checkpointInterval=4)
gbt.setCheckpointDir({ckpt!r})
import flowml.classification.gbt as gbtmod
# slow each round slightly so the parent can kill us mid-run
orig = gbtmod.grow_trees
def slow(*a, **k):
    import time; time.sleep(0.15)
    return orig(*a, **k)
gbtmod.grow_trees = slow
m = gbt.fit(df)
print("FULL_RUN_DONE", len(m._trees))
"""


def _data():
    r = np.random.default_rng(4)
    X = r.normal(size=(3000, 6)).astype(np.float32)
    y = ((X[:, 0] * X[:, 1] + X[:, 2]) > 0).astype(np.float64)
    return X, y


def test_kill_and_resume_reproduces_model(tmp_path):
    from flowml import FlowFrame
    from flowml.classification import GBTClassifier
    from flowml.utils.checkpoint import latest_gbt_checkpoint, load_gbt_checkpoint

    ckpt = str(tmp_path / "ck")
    script = _TRAIN_SCRIPT.format(repo=REPO, ckpt=ckpt)
    proc = subprocess.Popen(
        [sys.executable, "-c", script],
        stdout=subprocess.PIPE,
        stderr=subprocess.STDOUT,
    )
    # wait for at least one checkpoint, then SIGKILL mid-training
    deadline = time.time() + 120
    killed = False
    while time.time() < deadline:
        if proc.poll() is not None:
            break  # finished before we killed it (machine too fast) — fine
        latest = latest_gbt_checkpoint(ckpt)
        if latest is not None:
            proc.kill()
            killed = True
            break
        time.sleep(0.05)
    proc.wait(timeout=60)
    latest = latest_gbt_checkpoint(ckpt)
    assert latest is not None, "no checkpoint was written before the kill"

    # restart from the checkpoint
    init, rounds = load_gbt_checkpoint(latest)
    assert rounds % 4 == 0 and rounds >= 4
    X, y = _data()
    df = FlowFrame({"features": torch.from_numpy(X), "label": y})
    resumed = GBTClassifier(maxIter=16, maxDepth=3, seed=7)
    resumed.setInitModel(init)
    m_resumed = resumed.fit(df)

    # ground truth: uninterrupted run
    m_full = GBTClassifier(maxIter=16, maxDepth=3, seed=7).fit(df)
    p_full = m_full.transform(df)["rawPrediction"]
    p_res = m_resumed.transform(df)["rawPrediction"]
    assert torch.equal(p_full, p_res), "restart did not reproduce the model"
    assert len(m_resumed._trees) == len(m_full._trees) == 16


def test_checkpoint_atomicity(tmp_path):
    """A checkpoint dir is either absent or complete (tmp+rename)."""
    from flowml.utils.checkpoint import save_gbt_checkpoint, load_gbt_checkpoint
    from flowml.classification.tree import FlatTree

    tr = FlatTree(stat_width=3, regression=True)
    tr.add_node(np.array([1.0, 2.0, 4.0]), 0.5, 4.0)
    path = save_gbt_checkpoint(str(tmp_path / "ck"), [tr], [1.0], 4, 6)
    assert os.path.basename(path) == "round_000004"
    model, rounds = load_gbt_checkpoint(path)
    assert rounds == 4
    assert len(model._trees) == 1
