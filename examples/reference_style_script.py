"""A script written against the REFERENCE package import path — run it
unmodified with:  python -m spark_rapids_ml_amd examples/reference_style_script.py
"""
import numpy as np
from spark_rapids_ml.clustering import KMeans          # reference import path
from spark_rapids_ml.feature import PCA
from spark_rapids_ml_amd.data import DataFrame

X = np.random.default_rng(0).normal(size=(10_000, 32)).astype(np.float32)
df = DataFrame.from_numpy(X)
print("KMeans cost:", KMeans(k=8, maxIter=10).fit(df).trainingCost)
print("PCA variance ratios:", PCA(k=3).fit(df).explainedVariance)
