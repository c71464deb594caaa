"""flowml — MI355X-native network-traffic classification framework.

A from-scratch re-implementation of the capability surface of
``biagiom/spark-network-traffic-classifier`` (a Spark-ML pipeline app; see
SURVEY.md §1-§2 for the layer map): the ``pyspark.ml`` Pipeline API
(fit/transform), CSV flow-record ingestion, StringIndexer / OneHotEncoder /
VectorAssembler featurization, the classifiers the reference exposes
(LogisticRegression, RandomForestClassifier, MultilayerPerceptronClassifier,
plus GBTClassifier per BASELINE.json:10), the MulticlassClassificationEvaluator
and the Spark ML model-save format — rebuilt MI355X-first:

* columnar flow batches are torch tensors resident in HBM3E (``flowml.frame``),
* hot compute (per-feature histogram build, split-gain scan, binning, the
  MLP GEMM/ReLU/softmax stack, ensemble tree inference) runs in hand-written
  CDNA4 HIP kernels (``flowml/kernels/*.hip``, gfx950 MFMA + LDS),
* data-parallel training shards rows across GPUs, one process per GPU, and
  reduces histograms / gradients with RCCL all-reduce over xGMI
  (``flowml.dist``).

There is no JVM, no py4j, no lazy query planner, and no CUDA-compat layer.
"""

__version__ = "0.1.0"

from flowml.param import Param, Params
from flowml.base import Estimator, Transformer, Model
from flowml.pipeline import Pipeline, PipelineModel
from flowml.frame import FlowFrame, read_csv
from flowml import linalg
from flowml import classification, clustering, evaluation, feature, regression, tuning  # noqa: F401 (namespace re-exports)

__all__ = [
    "Param",
    "Params",
    "Estimator",
    "Transformer",
    "Model",
    "Pipeline",
    "PipelineModel",
    "FlowFrame",
    "read_csv",
    "linalg",
]
