"""spark_rapids_ml_amd — MI355X-native distributed classical ML.

A from-scratch AMD MI355X (CDNA4/gfx950) framework with the capabilities of
NVIDIA/spark-rapids-ml (reference: /root/reference): a pyspark.ml-compatible
Estimator/Model API over an SPMD runtime — one process per GPU, collectives via
torch.distributed (RCCL over xGMI on ROCm; gloo on CPU) — with hand-written
HIP kernels for the numeric hot paths (MFMA Gram/covariance, LDS-tiled fused
KMeans assignment, fused GLM gradients, tiled top-k for k-NN).

Unlike the reference (a Python glue layer over cuML/RAFT/cuVS wheels driven by
Spark barrier stages, reference core.py:742), this framework owns the whole
stack: API, distributed runtime, data plane and kernels. The user-facing
Estimator/Model/Param surface mirrors pyspark.ml so reference users find the
same classes, params, and persistence semantics.
"""

__version__ = "0.1.0"

from .models.clustering import KMeans, KMeansModel, DBSCAN, DBSCANModel
from .models.feature import PCA, PCAModel
from .models.regression import (
    LinearRegression,
    LinearRegressionModel,
    RandomForestRegressor,
    RandomForestRegressionModel,
)
from .models.classification import (
    LogisticRegression,
    LogisticRegressionModel,
    RandomForestClassifier,
    RandomForestClassificationModel,
)
from .models.knn import (
    NearestNeighbors,
    NearestNeighborsModel,
    ApproximateNearestNeighbors,
    ApproximateNearestNeighborsModel,
)
from .models.umap import UMAP, UMAPModel
from .tuning import CrossValidator, CrossValidatorModel
from .pipeline import Pipeline, PipelineModel

__all__ = [
    "KMeans",
    "KMeansModel",
    "DBSCAN",
    "DBSCANModel",
    "PCA",
    "PCAModel",
    "LinearRegression",
    "LinearRegressionModel",
    "RandomForestRegressor",
    "RandomForestRegressionModel",
    "LogisticRegression",
    "LogisticRegressionModel",
    "RandomForestClassifier",
    "RandomForestClassificationModel",
    "NearestNeighbors",
    "NearestNeighborsModel",
    "ApproximateNearestNeighbors",
    "ApproximateNearestNeighborsModel",
    "UMAP",
    "UMAPModel",
    "CrossValidator",
    "CrossValidatorModel",
    "Pipeline",
    "PipelineModel",
]
