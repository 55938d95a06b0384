"""Per-algorithm benchmarks (reference benchmark/bench_*.py, one class per
algorithm; dispatched by benchmark_runner.py). Dataset shapes default to the
reference's published workloads (1M×3000 f32) scaled by
--num_rows/--num_cols."""

from __future__ import annotations

import argparse
from typing import Dict, Optional

import numpy as np

from spark_rapids_ml_amd import (
    DBSCAN,
    KMeans,
    LinearRegression,
    LogisticRegression,
    NearestNeighbors,
    ApproximateNearestNeighbors,
    PCA,
    RandomForestClassifier,
    RandomForestRegressor,
    UMAP,
)
from spark_rapids_ml_amd.data import DataFrame

from .base import BenchmarkBase
from . import gen_data


class BenchKMeans(BenchmarkBase):
    name = "kmeans"

    def add_arguments(self, ap):
        ap.add_argument("--k", type=int, default=1000)
        ap.add_argument("--maxIter", type=int, default=30)
        ap.add_argument("--initMode", default="random")
        ap.add_argument("--tol", type=float, default=0.0)

    def make_data(self, args):
        X, _ = gen_data.gen_blobs(args.num_rows, args.num_cols, centers=args.k, seed=args.seed)
        return DataFrame.from_numpy(X)

    def make_estimator(self, args):
        return KMeans(k=args.k, maxIter=args.maxIter, initMode=args.initMode, tol=args.tol, seed=args.seed)

    def score(self, model, df, args):
        return {"inertia": model.trainingCost, "n_iter": model._model_attributes["n_iter_"]}


class BenchPCA(BenchmarkBase):
    name = "pca"

    def add_arguments(self, ap):
        ap.add_argument("--k", type=int, default=3)

    def make_data(self, args):
        X = gen_data.gen_low_rank_matrix(args.num_rows, args.num_cols, seed=args.seed)
        return DataFrame.from_numpy(X)

    def make_estimator(self, args):
        return PCA(k=args.k)

    def score(self, model, df, args):
        return {"explained_variance_ratio_sum": float(np.sum(model.explainedVariance))}


class BenchLinearRegression(BenchmarkBase):
    name = "linear_regression"

    def add_arguments(self, ap):
        ap.add_argument("--regParam", type=float, default=0.0)
        ap.add_argument("--elasticNetParam", type=float, default=0.0)
        ap.add_argument("--maxIter", type=int, default=100)

    def make_data(self, args):
        X, y = gen_data.gen_regression(args.num_rows, args.num_cols, seed=args.seed)
        return DataFrame.from_numpy(X, y)

    def make_estimator(self, args):
        return LinearRegression(
            regParam=args.regParam, elasticNetParam=args.elasticNetParam, maxIter=args.maxIter
        )

    def score(self, model, df, args):
        from spark_rapids_ml_amd.evaluation import RegressionEvaluator

        if "prediction" not in df.columns:
            return None
        return {"rmse": RegressionEvaluator(metricName="rmse").evaluate(df)}


class BenchLogisticRegression(BenchmarkBase):
    name = "logistic_regression"

    def add_arguments(self, ap):
        ap.add_argument("--regParam", type=float, default=1e-5)
        ap.add_argument("--maxIter", type=int, default=200)
        ap.add_argument("--n_classes", type=int, default=2)
        ap.add_argument("--sparse_density", type=float, default=None)

    def make_data(self, args):
        if args.sparse_density:
            nnz_row = max(1, int(args.sparse_density * args.num_cols))
            X, y = gen_data.gen_sparse_classification_fast(
                args.num_rows, args.num_cols, nnz_per_row=nnz_row, seed=args.seed
            )
            return DataFrame.from_numpy(X, y)
        X, y = gen_data.gen_classification(
            args.num_rows, args.num_cols, n_classes=args.n_classes, seed=args.seed
        )
        return DataFrame.from_numpy(X, y)

    def make_estimator(self, args):
        return LogisticRegression(regParam=args.regParam, maxIter=args.maxIter)

    def score(self, model, df, args):
        from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator

        if "prediction" not in df.columns:
            return None
        return {
            "accuracy": MulticlassClassificationEvaluator(metricName="accuracy").evaluate(df),
            "n_iter": model._model_attributes["n_iter_"],
        }


class BenchRandomForestClassifier(BenchmarkBase):
    name = "random_forest_classifier"

    def add_arguments(self, ap):
        ap.add_argument("--numTrees", type=int, default=50)
        ap.add_argument("--maxDepth", type=int, default=13)
        ap.add_argument("--maxBins", type=int, default=128)
        ap.add_argument("--n_classes", type=int, default=2)

    def make_data(self, args):
        X, y = gen_data.gen_classification(
            args.num_rows, args.num_cols, n_classes=args.n_classes, seed=args.seed
        )
        return DataFrame.from_numpy(X, y)

    def make_estimator(self, args):
        return RandomForestClassifier(
            numTrees=args.numTrees, maxDepth=args.maxDepth, maxBins=args.maxBins, seed=args.seed
        )

    def score(self, model, df, args):
        from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator

        if "prediction" not in df.columns:
            return None
        return {"accuracy": MulticlassClassificationEvaluator(metricName="accuracy").evaluate(df)}


class BenchRandomForestRegressor(BenchmarkBase):
    name = "random_forest_regressor"

    def add_arguments(self, ap):
        ap.add_argument("--numTrees", type=int, default=30)
        ap.add_argument("--maxDepth", type=int, default=6)
        ap.add_argument("--maxBins", type=int, default=128)

    def make_data(self, args):
        X, y = gen_data.gen_regression(args.num_rows, args.num_cols, seed=args.seed)
        return DataFrame.from_numpy(X, y)

    def make_estimator(self, args):
        return RandomForestRegressor(
            numTrees=args.numTrees, maxDepth=args.maxDepth, maxBins=args.maxBins, seed=args.seed
        )


class BenchNearestNeighbors(BenchmarkBase):
    name = "nearest_neighbors"

    def add_arguments(self, ap):
        ap.add_argument("--k", type=int, default=64)
        ap.add_argument("--num_queries", type=int, default=10000)

    def make_data(self, args):
        X, _ = gen_data.gen_blobs(args.num_rows, args.num_cols, seed=args.seed)
        return DataFrame.from_numpy(X)

    def make_estimator(self, args):
        return NearestNeighbors(k=args.k)

    def run_fit(self, est, df):
        model = est.fit(df)
        # the timed work is kneighbors (fit is a no-op; reference knn.py:347)
        q = df.take_local(np.arange(min(len(df), 10000)))
        model.kneighbors(q)
        return model


class BenchApproximateNearestNeighbors(BenchNearestNeighbors):
    name = "approximate_nearest_neighbors"

    def add_arguments(self, ap):
        super().add_arguments(ap)
        ap.add_argument("--algorithm", default="ivfflat")
        ap.add_argument("--nlist", type=int, default=256)
        ap.add_argument("--nprobe", type=int, default=32)

    def make_estimator(self, args):
        return ApproximateNearestNeighbors(
            k=args.k,
            algorithm=args.algorithm,
            algoParams={"nlist": args.nlist, "nprobe": args.nprobe},
        )


class BenchDBSCAN(BenchmarkBase):
    name = "dbscan"

    def add_arguments(self, ap):
        ap.add_argument("--eps", type=float, default=3.0)
        ap.add_argument("--min_samples", type=int, default=5)
        ap.add_argument("--algorithm", default="brute", choices=["brute", "rbc"])

    def make_data(self, args):
        X, _ = gen_data.gen_blobs(args.num_rows, args.num_cols, cluster_std=0.5, seed=args.seed)
        return DataFrame.from_numpy(X)

    def make_estimator(self, args):
        return DBSCAN(eps=args.eps, min_samples=args.min_samples,
                      algorithm=args.algorithm)


class BenchUMAP(BenchmarkBase):
    name = "umap"

    def add_arguments(self, ap):
        ap.add_argument("--n_neighbors", type=int, default=15)
        ap.add_argument("--n_epochs", type=int, default=200)
        ap.add_argument("--sample_fraction", type=float, default=1.0)

    def make_data(self, args):
        X, _ = gen_data.gen_blobs(args.num_rows, args.num_cols, seed=args.seed)
        return DataFrame.from_numpy(X)

    def make_estimator(self, args):
        return UMAP(
            n_neighbors=args.n_neighbors,
            n_epochs=args.n_epochs,
            sample_fraction=args.sample_fraction,
        )


BENCHMARKS = {
    b.name: b
    for b in [
        BenchKMeans(),
        BenchPCA(),
        BenchLinearRegression(),
        BenchLogisticRegression(),
        BenchRandomForestClassifier(),
        BenchRandomForestRegressor(),
        BenchNearestNeighbors(),
        BenchApproximateNearestNeighbors(),
        BenchDBSCAN(),
        BenchUMAP(),
    ]
}
