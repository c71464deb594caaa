"""Classifiers: LogisticRegression, RandomForest, GBT, MultilayerPerceptron.

The four classifier families of the reference app (BASELINE.json:5,10;
SURVEY.md §2.2), re-implemented MI355X-first: torch-ROCm tensors + hand
written CDNA4 HIP kernels on the hot paths, RCCL all-reduce for DP training.
"""

from flowml.classification.base import (
    Classifier,
    ClassificationModel,
    ProbabilisticClassifier,
    ProbabilisticClassificationModel,
)
from flowml.classification.logistic import LogisticRegression, LogisticRegressionModel
from flowml.classification.mlp import (
    MultilayerPerceptronClassifier,
    MultilayerPerceptronClassificationModel,
)
from flowml.classification.forest import (
    RandomForestClassifier,
    RandomForestClassificationModel,
    DecisionTreeClassifier,
    DecisionTreeClassificationModel,
)
from flowml.classification.gbt import GBTClassifier, GBTClassificationModel
from flowml.classification.naive_bayes import NaiveBayes, NaiveBayesModel
from flowml.classification.ovr import OneVsRest, OneVsRestModel
from flowml.classification.svc import LinearSVC, LinearSVCModel
from flowml.fm import FMClassifier, FMClassificationModel

__all__ = [
    "Classifier",
    "ClassificationModel",
    "ProbabilisticClassifier",
    "ProbabilisticClassificationModel",
    "LogisticRegression",
    "LogisticRegressionModel",
    "MultilayerPerceptronClassifier",
    "MultilayerPerceptronClassificationModel",
    "RandomForestClassifier",
    "RandomForestClassificationModel",
    "DecisionTreeClassifier",
    "DecisionTreeClassificationModel",
    "GBTClassifier",
    "GBTClassificationModel",
    "NaiveBayes",
    "NaiveBayesModel",
    "OneVsRest",
    "OneVsRestModel",
    "LinearSVC",
    "LinearSVCModel",
    "FMClassifier",
    "FMClassificationModel",
]
