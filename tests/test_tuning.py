"""ParamGridBuilder / CrossValidator / TrainValidationSplit +
BinaryClassificationEvaluator (pyspark.ml.tuning / .evaluation parity)."""

import numpy as np
import pytest
import torch
from sklearn import metrics as skm

from flowml import FlowFrame
from flowml.classification import LogisticRegression, RandomForestClassifier
from flowml.evaluation import (
    BinaryClassificationEvaluator,
    MulticlassClassificationEvaluator,
)
from flowml.tuning import (
    CrossValidator,
    ParamGridBuilder,
    TrainValidationSplit,
)


def _df(n=800, seed=0):
    r = np.random.default_rng(seed)
    X = r.normal(size=(n, 5)).astype(np.float32)
    y = (X[:, 0] + 0.5 * X[:, 1] + 0.3 * r.normal(size=n) > 0).astype(np.float64)
    return FlowFrame({"features": torch.from_numpy(X), "label": y})


def test_param_grid_builder():
    lr = LogisticRegression()
    grid = (
        ParamGridBuilder()
        .addGrid(lr.regParam, [0.0, 0.1])
        .addGrid(lr.maxIter, [5, 10, 20])
        .build()
    )
    assert len(grid) == 6
    assert {pm[lr.regParam] for pm in grid} == {0.0, 0.1}


def test_cross_validator_picks_sensible_model():
    df = _df()
    lr = LogisticRegression(maxIter=30)
    grid = ParamGridBuilder().addGrid(lr.regParam, [0.001, 10.0]).build()
    ev = MulticlassClassificationEvaluator(metricName="accuracy")
    cv = CrossValidator(estimator=lr, estimatorParamMaps=grid, evaluator=ev, numFolds=3, seed=1)
    m = cv.fit(df)
    assert len(m.avgMetrics) == 2
    # tiny regularization must beat the crushing one
    assert m.avgMetrics[0] > m.avgMetrics[1]
    assert m.bestModel.getRegParam() == 0.001
    out = m.transform(df)
    assert "prediction" in out


def test_train_validation_split():
    df = _df()
    rf = RandomForestClassifier(numTrees=5, seed=3)
    grid = ParamGridBuilder().addGrid(rf.maxDepth, [1, 5]).build()
    ev = MulticlassClassificationEvaluator(metricName="accuracy")
    tvs = TrainValidationSplit(estimator=rf, estimatorParamMaps=grid, evaluator=ev, seed=2)
    m = tvs.fit(df)
    assert len(m.validationMetrics) == 2
    assert m.bestModel.getMaxDepth() == 5


def test_validator_requires_config():
    with pytest.raises(ValueError):
        CrossValidator().fit(_df())


class TestBinaryEvaluator:
    def test_auc_matches_sklearn(self):
        r = np.random.default_rng(5)
        y = r.integers(0, 2, 500).astype(np.float64)
        score = y + r.normal(scale=0.9, size=500)
        raw = torch.stack([-torch.from_numpy(score), torch.from_numpy(score)], dim=1)
        df = FlowFrame({"label": y, "rawPrediction": raw})
        ev = BinaryClassificationEvaluator()
        assert ev.getMetricName() == "areaUnderROC"
        assert ev.evaluate(df) == pytest.approx(skm.roc_auc_score(y, score), abs=1e-6)

    def test_aupr_close_to_sklearn(self):
        r = np.random.default_rng(6)
        y = r.integers(0, 2, 500).astype(np.float64)
        score = y + r.normal(scale=0.9, size=500)
        df = FlowFrame({"label": y, "rawPrediction": torch.from_numpy(score)})
        ev = BinaryClassificationEvaluator(metricName="areaUnderPR")
        ours = ev.evaluate(df)
        theirs = skm.average_precision_score(y, score)
        assert ours == pytest.approx(theirs, abs=0.02)  # trapezoid vs step interp

    def test_on_fitted_model(self):
        df = _df()
        m = LogisticRegression(maxIter=30).fit(df)
        auc = BinaryClassificationEvaluator().evaluate(m.transform(df))
        assert auc > 0.9


class TestSmallerIsBetterMetric:
    def test_cv_with_rmse_picks_lower(self):
        """CrossValidator must respect isLargerBetter()=False (rmse)."""
        import numpy as np

        from flowml.evaluation import RegressionEvaluator
        from flowml.regression import LinearRegression
        from flowml.tuning import CrossValidator, ParamGridBuilder

        r = np.random.default_rng(0)
        X = torch.from_numpy(r.normal(size=(400, 4)).astype(np.float32))
        y = (X @ torch.tensor([1.0, -2.0, 0.5, 0.0]) + 0.05 *
             torch.from_numpy(r.normal(size=400).astype(np.float32))).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        lr = LinearRegression(maxIter=50)
        grid = (ParamGridBuilder()
                .addGrid(lr.regParam, [0.0, 50.0])  # huge reg must lose on rmse
                .build())
        cv = CrossValidator(estimator=lr, estimatorParamMaps=grid,
                            evaluator=RegressionEvaluator(metricName="rmse"),
                            numFolds=3, seed=1)
        m = cv.fit(df)
        assert m.avgMetrics[0] < m.avgMetrics[1]
        assert m.bestModel.getOrDefault(m.bestModel.regParam) == 0.0


class TestValidatorModelPersistence:
    def test_cv_model_roundtrip(self, tmp_path):
        import numpy as np

        from flowml.classification import LogisticRegression
        from flowml.evaluation import MulticlassClassificationEvaluator
        from flowml.tuning import CrossValidator, CrossValidatorModel, ParamGridBuilder

        r = np.random.default_rng(2)
        X = torch.from_numpy(r.normal(size=(300, 4)).astype(np.float32))
        y = (X[:, 0] > 0).to(torch.float64)
        df = FlowFrame({"features": X, "label": y})
        lr = LogisticRegression(maxIter=15)
        cv = CrossValidator(estimator=lr,
                            estimatorParamMaps=ParamGridBuilder().addGrid(lr.regParam, [0.0, 0.1]).build(),
                            evaluator=MulticlassClassificationEvaluator(metricName="accuracy"),
                            numFolds=2, seed=3)
        m = cv.fit(df)
        m.save(str(tmp_path / "cvm"))
        lm = CrossValidatorModel.load(str(tmp_path / "cvm"))
        assert lm.avgMetrics == pytest.approx(m.avgMetrics)
        assert torch.equal(lm.transform(df)["prediction"], m.transform(df)["prediction"])


def test_cross_validator_fold_col():
    from flowml.classification import LogisticRegression
    from flowml.evaluation import MulticlassClassificationEvaluator
    from flowml.tuning import CrossValidator, ParamGridBuilder

    r = np.random.default_rng(0)
    X = r.normal(size=(300, 4)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    folds = np.arange(300) % 3
    df = FlowFrame({"features": torch.from_numpy(X), "label": y,
                    "fold": folds.astype(np.float64)})
    lr = LogisticRegression(maxIter=10)
    grid = ParamGridBuilder().addGrid(lr.regParam, [0.0, 0.1]).build()
    cv = CrossValidator(estimator=lr, estimatorParamMaps=grid,
                        evaluator=MulticlassClassificationEvaluator(
                            metricName="accuracy"),
                        numFolds=3, foldCol="fold")
    m = cv.fit(df)
    assert len(m.avgMetrics) == 2 and max(m.avgMetrics) > 0.8
    bad = df.withColumn("fold", torch.full((300,), 5.0))
    with pytest.raises(ValueError, match="foldCol"):
        CrossValidator(estimator=lr, estimatorParamMaps=grid,
                       evaluator=MulticlassClassificationEvaluator(),
                       numFolds=3, foldCol="fold").fit(bad)
