#!/usr/bin/env python3
"""Train a network-traffic classifier on labeled flow CSVs — the flowml
equivalent of the reference app's driver script (load a
41-feature NSL-KDD-shaped CSV, featurize with StringIndexer + OneHotEncoder +
VectorAssembler, fit a classifier, evaluate accuracy/F1, save the model).

Single GPU / CPU:
    python examples/train_flows.py --data flows.csv --model rf --save out/model

Data-parallel over the node's GPUs (RCCL over xGMI, one process per GPU):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/train_flows.py --data flows.csv \
        --model rf --save out/model

With no --data, generates a synthetic NSL-KDD-shaped CSV first (no network
in the dev environment; see BASELINE.json "data": "synthetic").
"""

from __future__ import annotations

import argparse
import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch

from flowml import FlowFrame, Pipeline, dist, read_csv
from flowml.classification import (
    GBTClassifier,
    LogisticRegression,
    MultilayerPerceptronClassifier,
    RandomForestClassifier,
)
from flowml.evaluation import MulticlassClassificationEvaluator
from flowml.feature import OneHotEncoder, StringIndexer, VectorAssembler
from flowml.utils.config import RuntimeConfig


CATEGORICAL = ["protocol_type", "service", "flag"]


def synth_csv(path: str, rows: int = 20_000, seed: int = 7) -> None:
    r = np.random.default_rng(seed)
    protos = ["tcp", "udp", "icmp"]
    services = ["http", "smtp", "ftp", "dns", "ssh", "telnet"]
    flags = ["SF", "REJ", "S0", "RSTO"]
    labels = ["normal", "dos", "probe", "r2l", "u2r"]
    num = r.normal(size=(rows, 38))
    # ordered quantile classes of a noisy score over 5 informative features —
    # learnable by trees (axis-aligned) and by linear models alike
    score = num[:, :5] @ np.array([1.5, -1.2, 0.9, 1.1, -0.8]) + r.normal(scale=0.8, size=rows)
    qs = np.quantile(score, [0.45, 0.7, 0.85, 0.95])
    lbl_idx = np.searchsorted(qs, score)
    with open(path, "w") as f:
        cols = CATEGORICAL + [f"f{i}" for i in range(38)] + ["label"]
        f.write(",".join(cols) + "\n")
        for i in range(rows):
            row = [
                protos[int(r.integers(0, 3))],
                services[int(r.integers(0, 6))],
                flags[int(r.integers(0, 4))],
            ]
            row += [f"{v:.4f}" for v in num[i]]
            row.append(labels[lbl_idx[i]])
            f.write(",".join(row) + "\n")


def build_pipeline(model: str, numeric_cols, n_classes_hint: int = 5,
                   featurize: str = "onehot") -> Pipeline:
    indexer = StringIndexer(
        inputCols=CATEGORICAL,
        outputCols=[c + "_idx" for c in CATEGORICAL],
        handleInvalid="keep",
    )
    if featurize == "categorical" and model in ("rf", "gbt"):
        # tree path: skip one-hot — the indexed columns carry nominal
        # metadata, VectorAssembler marks those feature slots categorical,
        # and the trees use Spark categorical split-sets on them
        stages = [
            indexer,
            VectorAssembler(
                inputCols=[c + "_idx" for c in CATEGORICAL] + list(numeric_cols),
                outputCol="features",
            ),
            StringIndexer(inputCol="label", outputCol="labelIdx"),
        ]
    else:
        stages = [
            indexer,
            OneHotEncoder(
                inputCols=[c + "_idx" for c in CATEGORICAL],
                outputCols=[c + "_vec" for c in CATEGORICAL],
                handleInvalid="keep",
            ),
            VectorAssembler(
                inputCols=[c + "_vec" for c in CATEGORICAL] + list(numeric_cols),
                outputCol="features",
            ),
            StringIndexer(inputCol="label", outputCol="labelIdx"),
        ]
    clf_args = dict(featuresCol="features", labelCol="labelIdx")
    if model == "lr":
        clf = LogisticRegression(maxIter=100, **clf_args)
    elif model == "rf":
        clf = RandomForestClassifier(numTrees=100, maxDepth=8, **clf_args)
    elif model == "gbt":
        clf = GBTClassifier(maxIter=100, maxDepth=5, **clf_args)
    elif model == "mlp":
        # feature dim known after featurization; caller patches layers
        clf = MultilayerPerceptronClassifier(maxIter=100, **clf_args)
    else:
        raise SystemExit(f"unknown model {model}")
    stages.append(clf)
    return Pipeline(stages=stages)


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--data", default=None, help="labeled flow CSV (header row)")
    p.add_argument("--model", default="rf", choices=["lr", "rf", "gbt", "mlp"])
    p.add_argument("--save", default=None, help="save fitted PipelineModel here")
    p.add_argument("--test-fraction", type=float, default=0.2)
    p.add_argument("--seed", type=int, default=42)
    p.add_argument("--rows", type=int, default=20_000, help="synthetic rows if no --data")
    p.add_argument("--device", default=None, help="cpu | cuda:N (default: auto)")
    p.add_argument("--featurize", default="onehot", choices=["onehot", "categorical"],
                   help="categorical: trees split indexed columns with "
                        "Spark categorical split-sets instead of one-hot")
    args = p.parse_args()

    dist.init_from_env()
    cfg = RuntimeConfig()
    if args.device:
        cfg.device = args.device
    device = cfg.resolve_device()
    rank = dist.get_rank()

    data = args.data
    if data is None:
        data = os.path.join(tempfile.gettempdir(), f"flows_synth_{args.seed}.csv")
        if rank == 0 and not os.path.exists(data):
            synth_csv(data, rows=args.rows, seed=args.seed)
        dist.barrier()

    df = read_csv(data)
    if args.model == "gbt":  # GBT is binary (Spark parity): normal vs attack
        lab = df["label"]
        df = df.withColumn("label", np.where(lab == "normal", "normal", "attack"))
    # shard rows across ranks (contiguous static shard)
    lo, hi = dist.shard_range(df.count())
    df = df.take_rows(np.arange(lo, hi)).to(device)
    train, test = df.randomSplit([1 - args.test_fraction, args.test_fraction], seed=args.seed)

    numeric = [c for c in df.columns if c not in CATEGORICAL + ["label"]]
    pipe = build_pipeline(args.model, numeric, featurize=args.featurize)
    if args.model == "mlp":
        # resolve feature dim: fit featurization stages on a small head
        probe = Pipeline(stages=pipe.getStages()[:3]).fit(train.limit(512)).transform(train.limit(512))
        fdim = probe["features"].shape[1]
        n_classes = len(set(df["label"].tolist()))
        pipe.getStages()[-1].setLayers([fdim, 128, 64, n_classes])

    model = pipe.fit(train)
    pred = model.transform(test)
    # map predicted indices back to label strings (IndexToString)
    from flowml.feature import IndexToString

    label_stage = next(
        st for st in model.stages
        if type(st).__name__ == "StringIndexerModel" and st.getOrDefault(
            st.inputCol if st.isSet(st.inputCol) else st.inputCols
        ) == "label"
    )  # the label StringIndexerModel
    pred = IndexToString(
        inputCol="prediction", outputCol="predictedLabel", labels=label_stage.labels
    ).transform(pred)
    acc = MulticlassClassificationEvaluator(
        labelCol="labelIdx", metricName="accuracy"
    ).evaluate(pred)
    f1 = MulticlassClassificationEvaluator(labelCol="labelIdx", metricName="f1").evaluate(pred)
    mf1 = MulticlassClassificationEvaluator(
        labelCol="labelIdx", metricName="macroF1"
    ).evaluate(pred)
    if rank == 0:
        head = pred.select("predictedLabel").head(3)
        print(f"model={args.model} test accuracy={acc:.4f} weighted-F1={f1:.4f} macro-F1={mf1:.4f}")
        print("sample predictions:", [r["predictedLabel"] for r in head])
        if args.save:
            model.write().overwrite().save(args.save)
            print(f"saved PipelineModel to {args.save}")


if __name__ == "__main__":
    main()
