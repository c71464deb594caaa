"""The app-level driver (examples/train_flows.py) — the reference app's
equivalent — must run end-to-end on CPU for every model family."""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
CLI = os.path.join(REPO, "examples", "train_flows.py")


@pytest.mark.parametrize("model", ["lr", "rf", "gbt", "mlp"])
def test_cli_end_to_end(model, tmp_path):
    save = str(tmp_path / f"m_{model}")
    r = subprocess.run(
        [sys.executable, CLI, "--rows", "3000", "--model", model,
         "--save", save, "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r.returncode == 0, r.stdout[-800:] + r.stderr[-800:]
    assert "test accuracy=" in r.stdout
    assert os.path.isdir(os.path.join(save, "metadata"))
    assert os.path.isdir(os.path.join(save, "stages"))


def test_cli_categorical_featurization(tmp_path):
    """--featurize categorical: trees split the indexed columns with
    categorical split-sets (no one-hot stage in the pipeline)."""
    save = str(tmp_path / "m_rf_cat")
    r = subprocess.run(
        [sys.executable, CLI, "--rows", "3000", "--model", "rf",
         "--featurize", "categorical", "--save", save, "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r.returncode == 0, r.stdout[-800:] + r.stderr[-800:]
    assert "test accuracy=" in r.stdout
    # saved pipeline has 4 stages (indexer, assembler, label indexer, rf)
    stages = sorted(os.listdir(os.path.join(save, "stages")))
    assert len(stages) == 4


@pytest.mark.gpu
@pytest.mark.parametrize("featurize", ["onehot", "categorical"])
def test_cli_end_to_end_gpu(featurize, tmp_path):
    """The app driver end-to-end on a real GPU (HIP kernel path)."""
    save = str(tmp_path / f"m_gpu_{featurize}")
    r = subprocess.run(
        [sys.executable, CLI, "--rows", "20000", "--model", "rf",
         "--featurize", featurize, "--save", save, "--device", "cuda:0"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert r.returncode == 0, r.stdout[-800:] + r.stderr[-800:]
    assert "test accuracy=" in r.stdout
    assert os.path.isdir(os.path.join(save, "stages"))


def test_anomaly_example_cpu():
    cli = os.path.join(REPO, "examples", "detect_anomalies.py")
    r = subprocess.run(
        [sys.executable, cli, "--rows", "8000", "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r.returncode == 0, r.stdout[-500:] + r.stderr[-500:]
    line = [l for l in r.stdout.splitlines() if "precision" in l][0]
    prec = float(line.split("precision=")[1].split()[0])
    rec = float(line.split("recall=")[1].split()[0])
    assert prec > 0.9 and rec > 0.9, line


def test_serve_example_cpu(tmp_path):
    """train -> save -> load -> serve loop (examples/serve_flows.py)."""
    save = str(tmp_path / "m")
    r1 = subprocess.run(
        [sys.executable, CLI, "--rows", "3000", "--model", "rf",
         "--save", save, "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r1.returncode == 0, r1.stderr[-500:]
    serve = os.path.join(REPO, "examples", "serve_flows.py")
    r2 = subprocess.run(
        [sys.executable, serve, "--model-dir", save, "--batch", "1024",
         "--batches", "5", "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=REPO,
    )
    assert r2.returncode == 0, r2.stdout[-500:] + r2.stderr[-500:]
    assert "served 5120 flows" in r2.stdout
    assert "class counts:" in r2.stdout


@pytest.mark.gpu
def test_serve_example_gpu(tmp_path):
    save = str(tmp_path / "m")
    r1 = subprocess.run(
        [sys.executable, CLI, "--rows", "20000", "--model", "rf",
         "--save", save, "--device", "cuda:0"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert r1.returncode == 0, r1.stderr[-500:]
    serve = os.path.join(REPO, "examples", "serve_flows.py")
    r2 = subprocess.run(
        [sys.executable, serve, "--model-dir", save, "--batch", "65536",
         "--batches", "40", "--device", "cuda:0"],
        capture_output=True, text=True, timeout=600, cwd=REPO,
    )
    assert r2.returncode == 0, r2.stdout[-500:] + r2.stderr[-500:]
    assert "hipGraph=True" in r2.stdout
