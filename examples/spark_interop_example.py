"""Spark interop (round 2): fit on a pyspark DataFrame, proxy pyspark.ml,
save a stock-Spark-loadable model.

With pyspark installed:
    from spark_rapids_ml_amd.install import accelerate_pyspark
    accelerate_pyspark()                       # pyspark.ml.* now accelerated
    from pyspark.ml.clustering import KMeans   # resolves to the GPU class
    model = KMeans(k=8).fit(spark_df)          # Arrow bridge, SPMD ranks
    model.saveAsSparkModel("/models/km")       # stock Spark ML format

Without pyspark (this environment), the Spark-format persistence still
works — demonstrated below.
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from spark_rapids_ml_amd import KMeans, KMeansModel, LogisticRegression
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.spark import load_spark_model, spark_model_class

rng = np.random.default_rng(0)
X = rng.normal(size=(5000, 16)).astype(np.float32)
y = (X[:, 0] > 0).astype(np.float64)

km = KMeans(k=4, maxIter=10, seed=1).fit(DataFrame.from_numpy(X))
km.saveAsSparkModel("/tmp/km_spark_model", overwrite=True)
print("saved:", spark_model_class("/tmp/km_spark_model"))
# -> org.apache.spark.ml.clustering.KMeansModel; stock
#    pyspark.ml.clustering.KMeansModel.load("/tmp/km_spark_model") reads it

km2 = KMeansModel.loadFromSparkModel("/tmp/km_spark_model")
assert np.allclose(km2.cluster_centers_, km.cluster_centers_)

lr = LogisticRegression(maxIter=30).fit(DataFrame.from_numpy(X, y))
lr.saveAsSparkModel("/tmp/lr_spark_model", overwrite=True)
lr2 = load_spark_model("/tmp/lr_spark_model")
pred1 = np.asarray(lr.transform(DataFrame.from_numpy(X))["prediction"])
pred2 = np.asarray(lr2.transform(DataFrame.from_numpy(X))["prediction"])
assert (pred1 == pred2).all()
print("spark-format round trips OK")
