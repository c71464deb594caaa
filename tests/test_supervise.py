"""Restart supervisor (Spark task-retry analog)."""

import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
SUPERVISE = os.path.join(REPO, "scripts", "supervise.py")

# Child: fails until the marker file exists (created on first failure) —
# models a job that resumes successfully from its checkpoint on retry.
CHILD = r"""
import os, sys
marker = sys.argv[1]
if os.path.exists(marker):
    sys.exit(0)
open(marker, "w").close()
sys.exit(1)
"""


def test_restarts_until_success(tmp_path):
    marker = str(tmp_path / "marker")
    child = str(tmp_path / "child.py")
    with open(child, "w") as f:
        f.write(CHILD)
    r = subprocess.run(
        [sys.executable, SUPERVISE, "--max-failures", "4", "--backoff", "0", "--",
         sys.executable, child, marker],
        capture_output=True, text=True, timeout=60,
    )
    assert r.returncode == 0, r.stdout + r.stderr
    assert "attempt 2 succeeded" in r.stdout


def test_gives_up_after_max_failures(tmp_path):
    child = str(tmp_path / "child.py")
    with open(child, "w") as f:
        f.write("import sys; sys.exit(3)\n")
    r = subprocess.run(
        [sys.executable, SUPERVISE, "--max-failures", "2", "--backoff", "0", "--",
         sys.executable, child],
        capture_output=True, text=True, timeout=60,
    )
    assert r.returncode == 3
    assert "giving up after 3 attempts" in r.stdout
