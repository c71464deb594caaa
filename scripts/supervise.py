#!/usr/bin/env python3
"""Thin restart supervisor — the checkpoint-restart analog of Spark's task
retry / stage resubmission (``spark.task.maxFailures=4`` and
lineage recompute; reference `[S]`).

Single-node MI355X jobs have no lineage to replay, so recovery is
checkpoint-restart: the supervised command is expected to be idempotent given
its checkpoint directory (e.g. ``GBTClassifier.setCheckpointDir`` — see
``flowml/utils/checkpoint.py`` and ``tests/test_fault.py``). On a nonzero
exit the supervisor relaunches it, up to ``--max-failures`` times (default 4,
mirroring Spark's default); the relaunched run resumes from the newest
checkpoint and, because training is bitwise deterministic, reproduces the
exact model the uninterrupted run would have produced.

Usage:
    python scripts/supervise.py [--max-failures 4] [--backoff 1.0] -- \
        python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 train.py --checkpoint-dir ckpts/

Exit code: the final attempt's exit code (0 on success).
"""

from __future__ import annotations

import argparse
import subprocess
import sys
import time


def supervise(cmd, max_failures: int = 4, backoff: float = 1.0, _print=print) -> int:
    attempt = 0
    while True:
        attempt += 1
        _print(f"[supervise] attempt {attempt}: {' '.join(cmd)}", flush=True)
        rc = subprocess.call(cmd)
        if rc == 0:
            _print(f"[supervise] attempt {attempt} succeeded", flush=True)
            return 0
        _print(f"[supervise] attempt {attempt} exited with {rc}", flush=True)
        if attempt > max_failures:
            _print(f"[supervise] giving up after {attempt} attempts "
                   f"(max-failures={max_failures})", flush=True)
            return rc
        time.sleep(backoff)


def main(argv=None) -> int:
    p = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    p.add_argument("--max-failures", type=int, default=4,
                   help="retries allowed after the first failure (Spark's spark.task.maxFailures default)")
    p.add_argument("--backoff", type=float, default=1.0, help="seconds between attempts")
    p.add_argument("cmd", nargs=argparse.REMAINDER,
                   help="command to supervise (prefix with --)")
    args = p.parse_args(argv)
    cmd = args.cmd
    if cmd and cmd[0] == "--":
        cmd = cmd[1:]
    if not cmd:
        p.error("no command given (put it after --)")
    return supervise(cmd, max_failures=args.max_failures, backoff=args.backoff)


if __name__ == "__main__":
    sys.exit(main())
