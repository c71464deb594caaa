"""Golden-row featurizer tests (SURVEY.md §4 "MLTest.testTransformer")."""

import numpy as np
import pytest
import torch

from flowml import FlowFrame
from flowml.feature import (
    MinMaxScaler,
    OneHotEncoder,
    StandardScaler,
    StringIndexer,
    StringIndexerModel,
    VectorAssembler,
)


class TestStringIndexer:
    def test_frequency_desc_with_alpha_ties(self):
        # b:3, a:2, c:2 -> b=0, then tie a/c broken alphabetically: a=1, c=2
        df = FlowFrame({"s": ["b", "b", "b", "a", "a", "c", "c"]})
        m = StringIndexer(inputCol="s", outputCol="i").fit(df)
        assert m.labels == ["b", "a", "c"]
        out = m.transform(df)
        assert out["i"].tolist() == [0.0, 0.0, 0.0, 1.0, 1.0, 2.0, 2.0]
        assert out["i"].dtype == torch.float64

    def test_order_types(self):
        df = FlowFrame({"s": ["b", "b", "a", "c", "c", "c"]})
        cases = {
            "frequencyDesc": ["c", "b", "a"],
            "frequencyAsc": ["a", "b", "c"],
            "alphabetAsc": ["a", "b", "c"],
            "alphabetDesc": ["c", "b", "a"],
        }
        for order, expect in cases.items():
            m = StringIndexer(inputCol="s", outputCol="i", stringOrderType=order).fit(df)
            assert m.labels == expect, order

    def test_handle_invalid_error(self):
        train = FlowFrame({"s": ["a", "b"]})
        test = FlowFrame({"s": ["a", "zz"]})
        m = StringIndexer(inputCol="s", outputCol="i").fit(train)
        with pytest.raises(ValueError, match="unseen"):
            m.transform(test)

    def test_handle_invalid_skip_keep(self):
        train = FlowFrame({"s": ["a", "a", "b"]})
        test = FlowFrame({"s": ["a", "zz", "b"], "x": [1.0, 2.0, 3.0]})
        m_skip = StringIndexer(inputCol="s", outputCol="i", handleInvalid="skip").fit(train)
        out = m_skip.transform(test)
        assert out.count() == 2
        assert out["x"].tolist() == [1.0, 3.0]
        m_keep = StringIndexer(inputCol="s", outputCol="i", handleInvalid="keep").fit(train)
        out = m_keep.transform(test)
        assert out["i"].tolist() == [0.0, 2.0, 1.0]  # unseen -> numLabels

    def test_multi_col(self):
        df = FlowFrame({"s1": ["a", "b"], "s2": ["x", "x"]})
        m = StringIndexer(inputCols=["s1", "s2"], outputCols=["i1", "i2"]).fit(df)
        out = m.transform(df)
        assert "i1" in out and "i2" in out
        assert len(m.labelsArray) == 2

    def test_numeric_input_cast(self):
        df = FlowFrame({"n": [3.0, 3.0, 1.0]})
        m = StringIndexer(inputCol="n", outputCol="i").fit(df)
        assert m.labels == ["3.0", "1.0"]

    def test_from_labels(self):
        m = StringIndexerModel.from_labels(["x", "y"], inputCol="s", outputCol="i")
        out = m.transform(FlowFrame({"s": ["y", "x"]}))
        assert out["i"].tolist() == [1.0, 0.0]


class TestOneHotEncoder:
    def test_drop_last_default(self):
        df = FlowFrame({"i": [0.0, 1.0, 2.0, 1.0]})
        m = OneHotEncoder(inputCol="i", outputCol="v").fit(df)
        assert m.categorySizes == [3]
        out = m.transform(df)
        v = out["v"]
        assert v.shape == (4, 2)  # dropLast: 3 cats -> size 2
        assert v.tolist() == [[1, 0], [0, 1], [0, 0], [0, 1]]

    def test_no_drop_last(self):
        df = FlowFrame({"i": [0.0, 1.0, 2.0]})
        m = OneHotEncoder(inputCol="i", outputCol="v", dropLast=False).fit(df)
        assert m.transform(df)["v"].tolist() == [[1, 0, 0], [0, 1, 0], [0, 0, 1]]

    def test_handle_invalid(self):
        train = FlowFrame({"i": [0.0, 1.0]})
        test = FlowFrame({"i": [0.0, 5.0]})
        m = OneHotEncoder(inputCol="i", outputCol="v").fit(train)
        with pytest.raises(ValueError):
            m.transform(test)
        m_keep = OneHotEncoder(inputCol="i", outputCol="v", handleInvalid="keep").fit(train)
        v = m_keep.transform(test)["v"]
        # keep: size = numCats + 1 - dropLast = 2
        assert v.shape == (2, 2)
        assert v.tolist() == [[1, 0], [0, 0]]  # invalid -> extra cat (dropped? no: idx=2 of size 2 -> all-zero is last cat... idx 2 == size -> zero row)


class TestVectorAssembler:
    def test_concat_order(self):
        df = FlowFrame(
            {
                "a": [1.0, 2.0],
                "v": torch.tensor([[10.0, 20.0], [30.0, 40.0]]),
                "b": [5.0, 6.0],
            }
        )
        out = VectorAssembler(inputCols=["a", "v", "b"], outputCol="f").transform(df)
        assert out["f"].tolist() == [[1, 10, 20, 5], [2, 30, 40, 6]]
        assert out["f"].dtype == torch.float32

    def test_handle_invalid(self):
        df = FlowFrame({"a": [1.0, float("nan")]})
        va = VectorAssembler(inputCols=["a"], outputCol="f")
        with pytest.raises(ValueError):
            va.transform(df)
        out = va.copy({va.handleInvalid: "skip"}).transform(df)
        assert out.count() == 1
        out = va.copy({va.handleInvalid: "keep"}).transform(df)
        assert out.count() == 2

    def test_string_input_rejected(self):
        df = FlowFrame({"s": ["a", "b"]})
        with pytest.raises(TypeError):
            VectorAssembler(inputCols=["s"], outputCol="f").transform(df)


class TestScalers:
    def test_standard_scaler_defaults(self):
        x = torch.tensor([[1.0, 10.0], [2.0, 10.0], [3.0, 10.0]])
        df = FlowFrame({"v": x})
        m = StandardScaler(inputCol="v", outputCol="s").fit(df)
        out = m.transform(df)["s"]
        # withMean=False default: divide by sample std only
        std0 = float(x[:, 0].std(unbiased=True))
        assert torch.allclose(out[:, 0], x[:, 0] / std0, atol=1e-6)
        # zero-variance feature -> scaled to 0 (Spark: factor 0)
        assert out[:, 1].abs().sum() == 0

    def test_standard_scaler_with_mean(self):
        x = torch.tensor([[1.0], [3.0]])
        m = StandardScaler(inputCol="v", outputCol="s", withMean=True).fit(
            FlowFrame({"v": x})
        )
        out = m.transform(FlowFrame({"v": x}))["s"]
        assert abs(float(out.mean())) < 1e-6

    def test_minmax_scaler(self):
        x = torch.tensor([[0.0, 5.0], [10.0, 5.0]])
        m = MinMaxScaler(inputCol="v", outputCol="s").fit(FlowFrame({"v": x}))
        out = m.transform(FlowFrame({"v": x}))["s"]
        assert out[:, 0].tolist() == [0.0, 1.0]
        assert out[:, 1].tolist() == [0.5, 0.5]  # constant -> midpoint


class TestIndexToString:
    def test_round_trip_with_indexer(self):
        from flowml.feature import IndexToString

        df = FlowFrame({"s": ["b", "a", "b", "c"]})
        m = StringIndexer(inputCol="s", outputCol="i").fit(df)
        out = m.transform(df)
        back = IndexToString(inputCol="i", outputCol="s2", labels=m.labels).transform(out)
        assert back["s2"].tolist() == ["b", "a", "b", "c"]

    def test_out_of_range(self):
        from flowml.feature import IndexToString

        df = FlowFrame({"i": [0.0, 5.0]})
        with pytest.raises(ValueError):
            IndexToString(inputCol="i", outputCol="s", labels=["x"]).transform(df)


class TestNormalizer:
    def test_l2_default(self):
        from flowml.feature import Normalizer

        x = torch.tensor([[3.0, 4.0], [0.0, 0.0]])
        out = Normalizer(inputCol="v", outputCol="n").transform(FlowFrame({"v": x}))["n"]
        assert torch.allclose(out[0], torch.tensor([0.6, 0.8]))
        assert torch.equal(out[1], torch.zeros(2))  # zero row passes through

    def test_l1(self):
        from flowml.feature import Normalizer

        x = torch.tensor([[1.0, 3.0]])
        out = Normalizer(inputCol="v", outputCol="n", p=1.0).transform(FlowFrame({"v": x}))["n"]
        assert torch.allclose(out[0], torch.tensor([0.25, 0.75]))


class TestBinarizer:
    def test_scalar_and_vector(self):
        from flowml.feature import Binarizer

        df = FlowFrame({"a": [0.0, 0.5, 1.0], "v": torch.tensor([[-1.0, 2.0]] * 3)})
        b = Binarizer(inputCol="a", outputCol="ab", threshold=0.5)
        out = b.transform(df)
        assert out["ab"].tolist() == [0.0, 0.0, 1.0]  # strictly greater
        b2 = Binarizer(inputCol="v", outputCol="vb")
        assert b2.transform(df)["vb"].tolist() == [[0.0, 1.0]] * 3


class TestMaxAbsScaler:
    def test_fit_transform_roundtrip(self, tmp_path):
        from flowml.feature import MaxAbsScaler, MaxAbsScalerModel

        X = torch.tensor([[2.0, -4.0, 0.0], [-1.0, 2.0, 0.0]])
        df = FlowFrame({"features": X})
        m = MaxAbsScaler(inputCol="features", outputCol="scaled").fit(df)
        out = m.transform(df)["scaled"]
        assert torch.allclose(out, torch.tensor([[1.0, -1.0, 0.0], [-0.5, 0.5, 0.0]]))
        m.save(str(tmp_path / "mas"))
        lm = MaxAbsScalerModel.load(str(tmp_path / "mas"))
        assert torch.allclose(lm.transform(df)["scaled"], out)


class TestBucketizer:
    def test_buckets(self):
        from flowml.feature import Bucketizer

        b = Bucketizer(splits=[-float("inf"), 0.0, 10.0, float("inf")],
                       inputCol="x", outputCol="b")
        df = FlowFrame({"x": torch.tensor([-5.0, 0.0, 3.0, 10.0, 99.0])})
        out = b.transform(df)["b"].tolist()
        assert out == [0.0, 1.0, 1.0, 2.0, 2.0]

    def test_right_endpoint_closed_and_invalid(self):
        import pytest
        from flowml.feature import Bucketizer

        b = Bucketizer(splits=[0.0, 1.0, 2.0], inputCol="x", outputCol="b")
        df = FlowFrame({"x": torch.tensor([0.0, 1.9, 2.0])})
        assert b.transform(df)["b"].tolist() == [0.0, 1.0, 1.0]  # 2.0 in last
        bad = FlowFrame({"x": torch.tensor([-1.0, 0.5])})
        with pytest.raises(ValueError):
            b.transform(bad)
        b.setHandleInvalid("keep")
        assert b.transform(bad)["b"].tolist() == [2.0, 0.0]
        b.setHandleInvalid("skip")
        assert b.transform(bad)["b"].tolist() == [0.0]


class TestImputer:
    def test_mean_median_roundtrip(self, tmp_path):
        from flowml.feature import Imputer, ImputerModel

        nan = float("nan")
        df = FlowFrame({"a": torch.tensor([1.0, nan, 3.0]),
                        "b": torch.tensor([10.0, 20.0, nan])})
        m = Imputer(inputCols=["a", "b"], outputCols=["ai", "bi"]).fit(df)
        out = m.transform(df)
        assert out["ai"].tolist() == [1.0, 2.0, 3.0]
        assert out["bi"].tolist() == [10.0, 20.0, 15.0]
        med = Imputer(inputCols=["a"], outputCols=["ai"], strategy="median").fit(df)
        assert med.surrogates[0] == 1.0 or med.surrogates[0] == 3.0 or med.surrogates[0] == 2.0
        m.save(str(tmp_path / "imp"))
        lm = ImputerModel.load(str(tmp_path / "imp"))
        assert lm.surrogates == m.surrogates
        assert lm.transform(df)["ai"].tolist() == [1.0, 2.0, 3.0]


class TestQuantileDiscretizer:
    def test_equi_depth_buckets(self):
        from flowml.feature import QuantileDiscretizer

        x = torch.arange(100, dtype=torch.float32)
        df = FlowFrame({"x": x})
        model = QuantileDiscretizer(inputCol="x", outputCol="b", numBuckets=4).fit(df)
        out = model.transform(df)["b"]
        counts = torch.bincount(out.to(torch.int64))
        assert len(counts) == 4
        assert int(counts.min()) >= 20  # roughly equi-depth
        # fitted model is a plain Bucketizer (Spark returns Bucketizer too)
        from flowml.feature import Bucketizer
        assert isinstance(model, Bucketizer)

    def test_duplicate_quantiles_collapse(self):
        from flowml.feature import QuantileDiscretizer

        x = torch.tensor([1.0] * 50 + [2.0] * 50)
        df = FlowFrame({"x": x})
        m = QuantileDiscretizer(inputCol="x", outputCol="b", numBuckets=10).fit(df)
        assert len(m.getOrDefault(m.splits)) <= 5


class TestPCA:
    def test_matches_sklearn(self, tmp_path):
        from sklearn.decomposition import PCA as SkPCA

        from flowml.feature import PCA, PCAModel

        r = np.random.default_rng(5)
        base = r.normal(size=(500, 2))
        X = np.hstack([base, base @ r.normal(size=(2, 3))]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = PCA(k=2, inputCol="features", outputCol="p").fit(df)
        ours = m.transform(df)["p"].numpy()
        sk = SkPCA(n_components=2).fit(X)
        theirs = sk.transform(X)
        # components are sign/center ambiguous: compare |corr| per component
        for j in range(2):
            c = np.corrcoef(ours[:, j], theirs[:, j])[0, 1]
            assert abs(c) > 0.9999
        assert np.allclose(m.explainedVariance.toArray(),
                           sk.explained_variance_ratio_, atol=1e-4)
        m.save(str(tmp_path / "pca"))
        lm = PCAModel.load(str(tmp_path / "pca"))
        assert np.allclose(lm.transform(df)["p"].numpy(), ours)


class TestVectorSlicerAndInteraction:
    def test_slicer(self):
        from flowml.feature import VectorSlicer

        X = torch.arange(12, dtype=torch.float32).reshape(3, 4)
        df = FlowFrame({"v": X})
        out = VectorSlicer(inputCol="v", outputCol="s", indices=[3, 1]).transform(df)
        assert torch.equal(out["s"], X[:, [3, 1]])

    def test_interaction(self):
        from flowml.feature import Interaction

        a = torch.tensor([[1.0, 2.0], [3.0, 4.0]])
        b = torch.tensor([10.0, 100.0])
        df = FlowFrame({"a": a, "b": b})
        out = Interaction(inputCols=["a", "b"], outputCol="i").transform(df)
        assert torch.equal(out["i"], torch.tensor([[10.0, 20.0], [300.0, 400.0]]))
        # three-way: widths multiply
        c = torch.tensor([[1.0, -1.0], [2.0, 0.5]])
        df2 = FlowFrame({"a": a, "b": b, "c": c})
        out2 = Interaction(inputCols=["a", "b", "c"], outputCol="i").transform(df2)
        assert out2["i"].shape == (2, 4)
        assert torch.allclose(out2["i"][0], torch.tensor([10.0, -10.0, 20.0, -20.0]))


class TestStringOrderTypes:
    def test_all_four_orders(self):
        from flowml.feature import StringIndexer

        vals = ["b", "b", "b", "a", "c", "c"]  # freq: b=3, c=2, a=1
        df = FlowFrame({"s": np.array(vals, dtype=object)})

        def labels(order):
            return StringIndexer(inputCol="s", outputCol="i",
                                 stringOrderType=order).fit(df).labels

        assert labels("frequencyDesc") == ["b", "c", "a"]
        assert labels("frequencyAsc") == ["a", "c", "b"]
        assert labels("alphabetAsc") == ["a", "b", "c"]
        assert labels("alphabetDesc") == ["c", "b", "a"]


class TestRobustScaler:
    def test_fit_transform_quantiles(self):
        from flowml.feature import RobustScaler

        x = torch.tensor([[float(v)] for v in range(1, 10)])  # 1..9
        df = FlowFrame({"f": x})
        m = RobustScaler(inputCol="f", outputCol="s").fit(df)
        assert float(m.median[0]) == 5.0
        assert float(m.range[0]) == 4.0  # q75(7) - q25(3) torch linear interp
        out = m.transform(df)["s"]
        assert torch.allclose(out, x / 4.0)
        m2 = m.copy({m.withCentering: True})
        out2 = m2.transform(df)["s"]
        assert torch.allclose(out2, (x - 5.0) / 4.0)
        # scaling off
        m3 = m.copy({m.withCentering: True, m.withScaling: False})
        assert torch.allclose(m3.transform(df)["s"], x - 5.0)

    def test_zero_range(self):
        from flowml.feature import RobustScaler

        x = torch.ones(10, 2)
        m = RobustScaler(inputCol="f", outputCol="s").fit(FlowFrame({"f": x}))
        out = m.transform(FlowFrame({"f": x}))["s"]
        assert torch.all(out == 0)  # constant feature scales to 0

    def test_round_trip(self, tmp_path):
        from flowml.feature import RobustScaler, RobustScalerModel

        r = np.random.default_rng(3)
        x = torch.from_numpy(r.normal(size=(200, 4)).astype(np.float32))
        m = RobustScaler(inputCol="f", outputCol="s",
                         withCentering=True).fit(FlowFrame({"f": x}))
        p = str(tmp_path / "rs")
        m.save(p)
        m2 = RobustScalerModel.load(p)
        assert torch.allclose(m2.median, m.median)
        assert torch.allclose(m2.range, m.range)
        assert m2.getOrDefault(m2.withCentering) is True


class TestElementwiseAndPoly:
    def test_elementwise_product(self):
        from flowml.feature import ElementwiseProduct

        df = FlowFrame({"f": torch.tensor([[1.0, 2.0, 3.0]])})
        out = ElementwiseProduct(inputCol="f", outputCol="o",
                                 scalingVec=[0.0, 1.0, 2.0]).transform(df)
        assert out["o"].tolist() == [[0.0, 2.0, 6.0]]
        with pytest.raises(ValueError):
            ElementwiseProduct(inputCol="f", outputCol="o",
                               scalingVec=[1.0]).transform(df)

    def test_polynomial_expansion_spark_order(self):
        from flowml.feature import PolynomialExpansion

        df = FlowFrame({"f": torch.tensor([[2.0, 3.0]])})
        out = PolynomialExpansion(inputCol="f", outputCol="p", degree=3).transform(df)
        # the documented Spark sequence for (x, y) at degree 3
        assert out["p"].tolist() == [[2.0, 4.0, 8.0, 3.0, 6.0, 12.0,
                                      9.0, 18.0, 27.0]]


class TestFeatureSelectors:
    def _data(self):
        r = np.random.default_rng(9)
        X = r.normal(size=(500, 6)).astype(np.float32)
        y = r.integers(0, 3, 500).astype(np.float64)
        X[:, 2] += y
        X[:, 4] += 2 * y
        return X, y

    def test_anova_top_features(self):
        from flowml.feature import UnivariateFeatureSelector

        X, y = self._data()
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        m = UnivariateFeatureSelector(
            featuresCol="features", outputCol="sel", labelCol="label",
            featureType="continuous", labelType="categorical",
            selectionThreshold=2.0).fit(df)
        assert m.selectedFeatures == [2, 4]
        out = m.transform(df)["sel"]
        assert out.shape == (500, 2)
        assert torch.allclose(out[:, 0], torch.from_numpy(X[:, 2]))

    def test_modes(self):
        from flowml.feature import UnivariateFeatureSelector

        X, y = self._data()
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        base = dict(featuresCol="features", outputCol="sel", labelCol="label",
                    featureType="continuous", labelType="categorical")
        fpr = UnivariateFeatureSelector(selectionMode="fpr",
                                        selectionThreshold=1e-6, **base).fit(df)
        assert set(fpr.selectedFeatures) == {2, 4}
        fwe = UnivariateFeatureSelector(selectionMode="fwe",
                                        selectionThreshold=1e-5, **base).fit(df)
        assert set(fwe.selectedFeatures) == {2, 4}
        fdr = UnivariateFeatureSelector(selectionMode="fdr",
                                        selectionThreshold=1e-6, **base).fit(df)
        assert set(fdr.selectedFeatures) == {2, 4}
        pct = UnivariateFeatureSelector(selectionMode="percentile",
                                        selectionThreshold=0.34, **base).fit(df)
        assert len(pct.selectedFeatures) == 2

    def test_chisq_selector(self):
        from flowml.feature import ChiSqSelector

        r = np.random.default_rng(10)
        y = r.integers(0, 3, 600).astype(np.float64)
        Xc = r.integers(0, 4, size=(600, 4)).astype(np.float32)
        Xc[:, 1] = (y + r.integers(0, 2, 600)) % 4  # informative
        df = FlowFrame({"features": torch.from_numpy(Xc), "label": y})
        m = ChiSqSelector(featuresCol="features", outputCol="sel",
                          numTopFeatures=1).fit(df)
        assert m.selectedFeatures == [1]

    def test_round_trip(self, tmp_path):
        from flowml.feature import (UnivariateFeatureSelector,
                                    UnivariateFeatureSelectorModel)

        X, y = self._data()
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        m = UnivariateFeatureSelector(
            featuresCol="features", outputCol="sel", labelCol="label",
            featureType="continuous", labelType="categorical",
            selectionThreshold=2.0).fit(df)
        p = str(tmp_path / "sel")
        m.save(p)
        m2 = UnivariateFeatureSelectorModel.load(p)
        assert m2.selectedFeatures == [2, 4]
        assert torch.equal(m.transform(df)["sel"], m2.transform(df)["sel"])

    def test_unsupported_combination(self):
        from flowml.feature import UnivariateFeatureSelector

        X, y = self._data()
        df = FlowFrame({"features": torch.from_numpy(X), "label": y})
        with pytest.raises(ValueError, match="unsupported"):
            UnivariateFeatureSelector(
                featuresCol="features", outputCol="s", labelCol="label",
                featureType="categorical", labelType="continuous").fit(df)


class TestRFormula:
    def _df(self):
        r = np.random.default_rng(11)
        return FlowFrame({
            "clicked": np.array(r.choice(["no", "yes"], 40), dtype=object),
            "country": np.array(r.choice(["us", "ca", "mx"], 40), dtype=object),
            "hour": torch.from_numpy(r.normal(size=40).astype(np.float32)),
            "bytes": torch.from_numpy(r.normal(size=40).astype(np.float32)),
        })

    def test_dot_and_exclusion(self):
        from flowml.feature import RFormula

        df = self._df()
        m = RFormula(formula="clicked ~ . - bytes").fit(df)
        out = m.transform(df)
        # country one-hot (3 cats, dropLast -> 2) + hour
        assert out["features"].shape == (40, 3)
        assert set(out["label"].tolist()) <= {0.0, 1.0}
        # internal temp columns are dropped
        assert not [c for c in out.columns if c.startswith("__rf_")]

    def test_interaction_terms(self):
        from flowml.feature import RFormula

        df = self._df()
        m = RFormula(formula="clicked ~ hour:bytes").fit(df)
        out = m.transform(df)
        expect = df["hour"] * df["bytes"]
        assert torch.allclose(out["features"][:, 0], expect)
        with pytest.raises(ValueError, match="numeric"):
            RFormula(formula="clicked ~ hour:country").fit(df)

    def test_numeric_label_passthrough(self):
        from flowml.feature import RFormula

        df = self._df().withColumn("target", torch.arange(40.0))
        m = RFormula(formula="target ~ hour + bytes").fit(df)
        out = m.transform(df)
        assert torch.equal(out["label"], torch.arange(40.0))

    def test_trains_a_classifier_end_to_end(self):
        from flowml.classification import LogisticRegression
        from flowml.feature import RFormula
        from flowml.pipeline import Pipeline

        df = self._df()
        model = Pipeline(stages=[
            RFormula(formula="clicked ~ ."),
            LogisticRegression(maxIter=10),
        ]).fit(df)
        assert "prediction" in model.transform(df).columns

    def test_round_trip(self, tmp_path):
        from flowml.feature import RFormula, RFormulaModel

        df = self._df()
        m = RFormula(formula="clicked ~ country + hour:bytes").fit(df)
        p = str(tmp_path / "rf")
        m.save(p)
        m2 = RFormulaModel.load(p)
        a = m.transform(df)
        b = m2.transform(df)
        assert torch.equal(a["features"], b["features"])
        assert torch.equal(a["label"], b["label"])


class TestVarianceThresholdAndSizeHint:
    def test_variance_selector(self, tmp_path):
        from flowml.feature import (VarianceThresholdSelector,
                                    VarianceThresholdSelectorModel)

        r = np.random.default_rng(0)
        X = np.column_stack([
            r.normal(size=200),            # var ~1
            np.full(200, 3.0),             # constant -> dropped at default 0
            r.normal(scale=0.01, size=200) # tiny var
        ]).astype(np.float32)
        df = FlowFrame({"features": torch.from_numpy(X)})
        m = VarianceThresholdSelector(outputCol="s").fit(df)
        assert m.selectedFeatures == [0, 2]
        m2 = VarianceThresholdSelector(outputCol="s",
                                       varianceThreshold=0.01).fit(df)
        assert m2.selectedFeatures == [0]
        p = str(tmp_path / "vts")
        m.save(p)
        m3 = VarianceThresholdSelectorModel.load(p)
        assert m3.selectedFeatures == [0, 2]
        assert m3.transform(df)["s"].shape == (200, 2)

    def test_vector_size_hint(self):
        from flowml.feature import VectorSizeHint

        df = FlowFrame({"f": torch.zeros(5, 3)})
        out = VectorSizeHint(inputCol="f", size=3).transform(df)
        assert out.meta("f")["size"] == 3
        with pytest.raises(ValueError, match="vector size"):
            VectorSizeHint(inputCol="f", size=4).transform(df)
