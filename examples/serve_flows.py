#!/usr/bin/env python3
"""Serve a saved flow-classification PipelineModel over micro-batches.

The second half of the reference app's lifecycle: train_flows.py fits and
saves; this loads the Spark-format model directory and classifies a stream
of flow CSVs/batches with the hipGraph-captured scorer (BASELINE config 5).

    python examples/train_flows.py --model rf --save /tmp/flowmodel
    python examples/serve_flows.py --model-dir /tmp/flowmodel \
        [--batch 262144] [--batches 100] [--device cuda:0]

With no --data, replays synthetic NSL-KDD-shaped batches. Multi-GPU:
launch under torchrun; ranks round-robin the stream (StreamingDriver) and
the aggregate throughput is all-reduced.
"""

from __future__ import annotations

import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from flowml import FlowFrame, dist
from flowml.io import load
from flowml.streaming import StreamingClassifier, StreamingDriver
from flowml.utils.config import RuntimeConfig


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model-dir", required=True,
                   help="saved PipelineModel (train_flows.py --save)")
    p.add_argument("--batch", type=int, default=262144)
    p.add_argument("--batches", type=int, default=50)
    p.add_argument("--device", default=None)
    p.add_argument("--seed", type=int, default=9)
    args = p.parse_args()

    dist.init_from_env()
    cfg = RuntimeConfig()
    if args.device:
        cfg.device = args.device
    device = torch.device(cfg.resolve_device())
    rank, world = dist.get_rank(), dist.get_world_size()

    pm = load(args.model_dir)
    # the featurization stages up to the classifier run per batch; the
    # classifier stage feeds the hipGraph scorer
    stages = list(pm.stages)
    clf = stages[-1]
    feats = stages[:-1]

    r = np.random.default_rng(args.seed + rank)

    def _stage_inputs(st):
        cols = []
        for pname in ("inputCol", "inputCols"):
            p2 = getattr(st, pname, None)
            if p2 is not None and (st.isSet(p2) or st.hasDefault(p2)):
                v = st.getOrDefault(p2)
                cols += [v] if isinstance(v, str) else list(v or [])
        return cols

    def featurize(raw: FlowFrame) -> torch.Tensor:
        df = raw
        for st in feats:
            ins = _stage_inputs(st)
            if ins and any(c not in df.columns for c in ins):
                continue  # label-indexing stages have no input at serve time
            df = st.transform(df)
        return df.col(clf.getFeaturesCol())

    # synthetic raw batches shaped like train_flows.py's CSV
    protos = np.array(["tcp", "udp", "icmp"], dtype=object)
    services = np.array(["http", "smtp", "ftp", "dns", "ssh", "telnet"],
                        dtype=object)
    flags = np.array(["SF", "REJ", "S0", "RSTO"], dtype=object)

    def raw_batch():
        n = args.batch
        cols = {
            "protocol_type": protos[r.integers(0, 3, n)],
            "service": services[r.integers(0, 6, n)],
            "flag": flags[r.integers(0, 4, n)],
        }
        for i in range(38):
            cols[f"f{i}"] = torch.from_numpy(
                r.normal(size=n).astype(np.float32))
        return FlowFrame(cols, device=device)

    # one featurized probe batch sizes the scorer; string stages run on
    # host per batch (they are not capture-safe), the numeric tail + the
    # ensemble forward replay as a hipGraph
    probe = featurize(raw_batch())
    sc = StreamingClassifier(clf, batch_size=args.batch,
                             num_features=probe.shape[1], device=device,
                             use_graph=device.type == "cuda")
    driver = StreamingDriver(sc)

    pool = [featurize(raw_batch()) for _ in range(2)]
    res = driver.run((pool[i % 2] for i in range(args.batches * world)),
                     count_classes=True)
    if rank == 0:
        top = int(np.argmax(res["class_counts"]))
        print(f"served {res['rows']} flows in {res['elapsed_s']:.2f}s = "
              f"{res['flows_per_s']/1e6:.3f}M flows/s "
              f"(world={world}, hipGraph={sc._graph is not None})")
        print(f"class counts: {[int(c) for c in res['class_counts']]} "
              f"(modal class index {top})")


if __name__ == "__main__":
    main()
