"""Pipeline fit/transform semantics."""

import numpy as np
import pytest
import torch

from flowml import FlowFrame, Pipeline
from flowml.base import Transformer
from flowml.classification import LogisticRegression
from flowml.feature import StringIndexer, VectorAssembler
from flowml.pipeline import PipelineModel


def _df():
    r = np.random.default_rng(0)
    n = 200
    return FlowFrame(
        {
            "s": r.choice(["a", "b"], size=n),
            "x": r.normal(size=n),
            "label": r.integers(0, 2, n).astype(np.float64),
        }
    )


def test_pipeline_fit_transform():
    df = _df()
    pipe = Pipeline(
        stages=[
            StringIndexer(inputCol="s", outputCol="si"),
            VectorAssembler(inputCols=["si", "x"], outputCol="features"),
            LogisticRegression(maxIter=5),
        ]
    )
    model = pipe.fit(df)
    assert isinstance(model, PipelineModel)
    assert len(model.stages) == 3
    out = model.transform(df)
    for c in ("si", "features", "rawPrediction", "probability", "prediction"):
        assert c in out


def test_transformer_only_pipeline():
    df = _df()
    pipe = Pipeline(stages=[StringIndexer(inputCol="s", outputCol="si")])
    model = pipe.fit(df)
    assert len(model.stages) == 1


def test_trailing_transformer_not_fitted_during_fit():
    """Stages after the last estimator are collected but NOT applied in fit."""

    calls = []

    class Probe(Transformer):
        def _transform(self, dataset):
            calls.append(1)
            return dataset

    df = _df()
    pipe = Pipeline(
        stages=[
            StringIndexer(inputCol="s", outputCol="si"),
            Probe(),
        ]
    )
    model = pipe.fit(df)
    assert calls == []  # trailing transformer untouched during fit
    model.transform(df)
    assert calls == [1]


def test_invalid_stage_type():
    with pytest.raises(TypeError):
        Pipeline(stages=["not a stage"]).fit(_df())


def test_pipeline_copy_deep_stages():
    si = StringIndexer(inputCol="s", outputCol="si")
    p = Pipeline(stages=[si])
    c = p.copy()
    assert c.getStages()[0] is not si
    assert c.getStages()[0].getInputCol() == "s"


class TestPipelinesWithNewFamilies:
    def test_kmeans_in_pipeline(self):
        import numpy as np

        from flowml.clustering import KMeans
        from flowml.feature import StandardScaler, VectorAssembler

        r = np.random.default_rng(0)
        a = torch.from_numpy(r.normal(size=300).astype(np.float32))
        b = torch.from_numpy((r.normal(size=300) + np.repeat([0, 8], 150)).astype(np.float32))
        df = FlowFrame({"a": a, "b": b})
        pm = Pipeline(stages=[
            VectorAssembler(inputCols=["a", "b"], outputCol="raw"),
            StandardScaler(inputCol="raw", outputCol="features"),
            KMeans(k=2, seed=3),
        ]).fit(df)
        pred = pm.transform(df)["prediction"]
        # the two b-shifted halves should split into the two clusters
        first = pred[:150]
        assert float((first == first[0]).float().mean()) > 0.95

    def test_ovr_in_pipeline_with_save(self, tmp_path):
        import numpy as np

        from flowml.classification import LinearSVC, OneVsRest
        from flowml.feature import VectorAssembler

        r = np.random.default_rng(1)
        x1 = torch.from_numpy(r.normal(size=400).astype(np.float32))
        x2 = torch.from_numpy(r.normal(size=400).astype(np.float32))
        y = torch.from_numpy(r.integers(0, 3, 400).astype(np.float64))
        x1 = x1 + (y == 1).float() * 3
        x2 = x2 - (y == 2).float() * 3
        df = FlowFrame({"x1": x1, "x2": x2, "label": y})
        pm = Pipeline(stages=[
            VectorAssembler(inputCols=["x1", "x2"], outputCol="features"),
            OneVsRest(classifier=LinearSVC(maxIter=30)),
        ]).fit(df)
        pred = pm.transform(df)["prediction"]
        acc = float((pred == y).float().mean())
        assert acc > 0.85
