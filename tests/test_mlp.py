"""MultilayerPerceptronClassifier tests."""

import numpy as np
import pytest
import torch

from flowml import FlowFrame
from flowml.classification import MultilayerPerceptronClassifier
from flowml.classification.mlp import _flatten, _unflatten, n_weights
from flowml.evaluation import MulticlassClassificationEvaluator


def _df(n=1200, f=6, k=3, seed=5):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, f)).astype(np.float32)
    W = r.normal(size=(f, k))
    y = (X @ W).argmax(1).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y})


def test_weight_layout_round_trip():
    layers = [4, 7, 3]
    flat = torch.arange(n_weights(layers), dtype=torch.float32)
    Ws, bs = _unflatten(flat, layers)
    assert Ws[0].shape == (7, 4)
    assert bs[0].shape == (7,)
    assert Ws[1].shape == (3, 7)
    back = _flatten(Ws, bs)
    assert torch.equal(back, flat)


def test_fit_accuracy():
    df = _df()
    m = MultilayerPerceptronClassifier(layers=[6, 16, 3], maxIter=80, seed=1).fit(df)
    acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(m.transform(df))
    assert acc > 0.9


def test_deterministic_given_seed():
    df = _df(n=400)
    m1 = MultilayerPerceptronClassifier(layers=[6, 8, 3], maxIter=10, seed=3).fit(df)
    m2 = MultilayerPerceptronClassifier(layers=[6, 8, 3], maxIter=10, seed=3).fit(df)
    assert torch.equal(m1._weights, m2._weights)


def test_layers_validation():
    df = _df(n=100)
    with pytest.raises(ValueError):
        MultilayerPerceptronClassifier(layers=[6], maxIter=2).fit(df)
    with pytest.raises(ValueError):
        MultilayerPerceptronClassifier(layers=[5, 4, 3], maxIter=2).fit(df)  # wrong input dim


def test_gd_solver():
    df = _df(n=400)
    m = MultilayerPerceptronClassifier(
        layers=[6, 8, 3], maxIter=100, solver="gd", stepSize=0.5, seed=1
    ).fit(df)
    acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(m.transform(df))
    assert acc > 0.6


def test_sigmoid_activation_parity_mode():
    df = _df(n=400)
    m = MultilayerPerceptronClassifier(
        layers=[6, 8, 3], maxIter=40, activation="sigmoid", seed=1
    ).fit(df)
    acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(m.transform(df))
    assert acc > 0.7


def test_probability_normalized():
    df = _df(n=200)
    m = MultilayerPerceptronClassifier(layers=[6, 8, 3], maxIter=10, seed=1).fit(df)
    prob = m.transform(df)["probability"]
    assert torch.allclose(prob.sum(dim=1), torch.ones(len(df)), atol=1e-5)
