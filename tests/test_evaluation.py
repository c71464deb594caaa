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


class TestExtendedMetrics:
    def _df(self):
        import torch
        from flowml.frame import FlowFrame

        pred = torch.tensor([0.0, 1.0, 2.0, 0.0, 1.0, 2.0, 0.0, 1.0])
        label = torch.tensor([0.0, 1.0, 2.0, 1.0, 0.0, 2.0, 0.0, 2.0])
        prob = torch.tensor([
            [0.7, 0.2, 0.1], [0.1, 0.8, 0.1], [0.2, 0.2, 0.6], [0.5, 0.3, 0.2],
            [0.3, 0.6, 0.1], [0.1, 0.1, 0.8], [0.9, 0.05, 0.05], [0.2, 0.5, 0.3],
        ])
        return FlowFrame({"prediction": pred, "label": label, "probability": prob})

    def test_hamming_loss(self):
        from flowml.evaluation import MulticlassClassificationEvaluator

        ev = MulticlassClassificationEvaluator(metricName="hammingLoss")
        df = self._df()
        acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(df)
        assert abs(ev.evaluate(df) - (1 - acc)) < 1e-12
        assert not ev.isLargerBetter()

    def test_log_loss(self):
        import numpy as np
        from flowml.evaluation import MulticlassClassificationEvaluator

        df = self._df()
        ev = MulticlassClassificationEvaluator(metricName="logLoss")
        p = df.col("probability").numpy().astype(np.float64)
        y = df.col("label").numpy().astype(int)
        expect = -np.log(np.clip(p[np.arange(len(y)), y], 1e-15, None)).mean()
        assert abs(ev.evaluate(df) - expect) < 1e-12
        assert not ev.isLargerBetter()

    def test_weighted_tpr_equals_weighted_recall(self):
        from flowml.evaluation import MulticlassClassificationEvaluator

        df = self._df()
        a = MulticlassClassificationEvaluator(metricName="weightedTruePositiveRate").evaluate(df)
        b = MulticlassClassificationEvaluator(metricName="weightedRecall").evaluate(df)
        assert a == b

    def test_weighted_fpr_range(self):
        from flowml.evaluation import MulticlassClassificationEvaluator

        v = MulticlassClassificationEvaluator(metricName="weightedFalsePositiveRate").evaluate(self._df())
        assert 0.0 <= v <= 1.0


class TestTrainingSummaries:
    def _df(self):
        import numpy as np
        import torch
        from flowml.frame import FlowFrame

        r = np.random.default_rng(0)
        X = torch.from_numpy(r.normal(size=(400, 5)).astype(np.float32))
        y = (X[:, 0] + 0.5 * X[:, 1] > 0).to(torch.float64)
        return FlowFrame({"features": X, "label": y})

    def test_rf_summary(self):
        from flowml.classification import RandomForestClassifier

        df = self._df()
        m = RandomForestClassifier(numTrees=5, maxDepth=3, seed=1).fit(df)
        assert m.hasSummary()
        s = m.summary
        assert 0.5 < s.accuracy <= 1.0
        assert s.labels == [0.0, 1.0]
        assert len(s.precisionByLabel) == 2
        assert 0.0 <= s.weightedFalsePositiveRate <= 1.0
        assert abs(s.weightedTruePositiveRate - s.weightedRecall) < 1e-12
        assert s.predictions.count() == df.count()

    def test_lr_summary_metrics(self):
        from flowml.classification import LogisticRegression

        m = LogisticRegression(maxIter=20).fit(self._df())
        s = m.summary
        assert s.totalIterations >= 1
        assert s.objectiveHistory[0] >= s.objectiveHistory[-1]
        assert 0.5 < s.accuracy <= 1.0
        assert len(s.fMeasureByLabel()) == 2

    def test_mlp_gbt_summary(self):
        from flowml.classification import GBTClassifier, MultilayerPerceptronClassifier

        df = self._df()
        g = GBTClassifier(maxIter=3, maxDepth=2, seed=0).fit(df)
        assert g.hasSummary() and 0.5 < g.summary.accuracy <= 1.0
        p = MultilayerPerceptronClassifier(layers=[5, 8, 2], maxIter=10, seed=0).fit(df)
        assert p.hasSummary() and 0.0 < p.summary.accuracy <= 1.0

    def test_loaded_model_has_no_summary(self, tmp_path):
        import pytest
        from flowml.classification import RandomForestClassifier, RandomForestClassificationModel

        m = RandomForestClassifier(numTrees=3, maxDepth=2, seed=1).fit(self._df())
        m.save(str(tmp_path / "m"))
        loaded = RandomForestClassificationModel.load(str(tmp_path / "m"))
        assert not loaded.hasSummary()
        with pytest.raises(RuntimeError):
            _ = loaded.summary


class TestThresholds:
    def test_multiclass_thresholds(self):
        import numpy as np
        from flowml.classification import RandomForestClassifier
        from flowml.frame import FlowFrame
        import torch

        r = np.random.default_rng(1)
        X = torch.from_numpy(r.normal(size=(300, 4)).astype(np.float32))
        y = (X[:, 0] > 0).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        m = RandomForestClassifier(numTrees=5, maxDepth=3, seed=0).fit(df)
        base = m.transform(df)["prediction"]
        # huge threshold on class 1 -> class 1 nearly never predicted
        m.setThresholds([1e-6, 1.0])
        t1 = m.transform(df)["prediction"]
        assert float((t1 == 1).sum()) <= float((base == 1).sum())
        assert float((t1 == 0).sum()) > float((base == 0).sum()) * 0.9
        # zero threshold wins outright
        m.setThresholds([0.0, 1.0])
        t0 = m.transform(df)["prediction"]
        assert torch.all(t0 == 0)
        # wrong length raises
        m.setThresholds([0.5, 0.5, 0.5])
        import pytest
        with pytest.raises(ValueError):
            m.transform(df)


class TestWeightedEvaluators:
    def test_weighted_accuracy_equals_duplication(self):
        import torch

        from flowml.evaluation import MulticlassClassificationEvaluator

        pred = torch.tensor([0.0, 1.0, 1.0, 0.0])
        lab = torch.tensor([0.0, 1.0, 0.0, 0.0])
        w = torch.tensor([3.0, 1.0, 2.0, 1.0])
        df_w = FlowFrame({"prediction": pred, "label": lab, "w": w})
        idx = torch.repeat_interleave(torch.arange(4), w.to(torch.int64))
        df_d = FlowFrame({"prediction": pred[idx], "label": lab[idx]})
        ev_w = MulticlassClassificationEvaluator(metricName="accuracy", weightCol="w")
        ev = MulticlassClassificationEvaluator(metricName="accuracy")
        assert ev_w.evaluate(df_w) == pytest.approx(ev.evaluate(df_d))
        f1_w = MulticlassClassificationEvaluator(metricName="f1", weightCol="w").evaluate(df_w)
        f1_d = MulticlassClassificationEvaluator(metricName="f1").evaluate(df_d)
        assert f1_w == pytest.approx(f1_d)

    def test_weighted_rmse_equals_duplication(self):
        import torch

        from flowml.evaluation import RegressionEvaluator

        pred = torch.tensor([1.0, 2.0, 4.0])
        lab = torch.tensor([1.5, 2.0, 3.0])
        w = torch.tensor([2.0, 1.0, 3.0])
        df_w = FlowFrame({"prediction": pred, "label": lab, "w": w})
        idx = torch.repeat_interleave(torch.arange(3), w.to(torch.int64))
        df_d = FlowFrame({"prediction": pred[idx], "label": lab[idx]})
        a = RegressionEvaluator(metricName="rmse", weightCol="w").evaluate(df_w)
        b = RegressionEvaluator(metricName="rmse").evaluate(df_d)
        assert a == pytest.approx(b)
