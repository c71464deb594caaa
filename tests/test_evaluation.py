"""MulticlassClassificationEvaluator vs sklearn oracle."""

import numpy as np
import pytest
import torch
from sklearn import metrics as skm

from flowml import FlowFrame
from flowml.evaluation import MulticlassClassificationEvaluator


@pytest.fixture
def preds(rng):
    y = rng.integers(0, 5, 500).astype(np.float64)
    p = y.copy()
    flip = rng.random(500) < 0.3
    p[flip] = rng.integers(0, 5, int(flip.sum())).astype(np.float64)
    return FlowFrame({"label": y, "prediction": p}), y, p


def test_accuracy(preds):
    df, y, p = preds
    ev = MulticlassClassificationEvaluator(metricName="accuracy")
    assert ev.evaluate(df) == pytest.approx(skm.accuracy_score(y, p))


def test_weighted_f1_default(preds):
    df, y, p = preds
    ev = MulticlassClassificationEvaluator()
    assert ev.getMetricName() == "f1"
    assert ev.evaluate(df) == pytest.approx(skm.f1_score(y, p, average="weighted"))


def test_macro_f1(preds):
    df, y, p = preds
    ev = MulticlassClassificationEvaluator(metricName="macroF1")
    assert ev.evaluate(df) == pytest.approx(skm.f1_score(y, p, average="macro"))


def test_weighted_precision_recall(preds):
    df, y, p = preds
    evp = MulticlassClassificationEvaluator(metricName="weightedPrecision")
    evr = MulticlassClassificationEvaluator(metricName="weightedRecall")
    assert evp.evaluate(df) == pytest.approx(skm.precision_score(y, p, average="weighted", zero_division=0))
    assert evr.evaluate(df) == pytest.approx(skm.recall_score(y, p, average="weighted", zero_division=0))


def test_by_label(preds):
    df, y, p = preds
    for lbl in range(5):
        ev = MulticlassClassificationEvaluator(metricName="precisionByLabel", metricLabel=float(lbl))
        assert ev.evaluate(df) == pytest.approx(
            skm.precision_score(y, p, labels=[lbl], average="macro", zero_division=0)
        )


def test_weighted_rows():
    df = FlowFrame(
        {
            "label": [0.0, 0.0, 1.0],
            "prediction": [0.0, 1.0, 1.0],
            "w": [2.0, 1.0, 1.0],
        }
    )
    ev = MulticlassClassificationEvaluator(metricName="accuracy", weightCol="w")
    assert ev.evaluate(df) == pytest.approx(3.0 / 4.0)


def test_is_larger_better():
    assert MulticlassClassificationEvaluator().isLargerBetter()
