"""RandomForest classification + regression: ensemble-parallel fit (trees
split across ranks), feature importances, single-vector predict, persistence.

Single process:  python examples/random_forest_example.py
Multi-GPU:       srml-amd-launch examples/random_forest_example.py
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from spark_rapids_ml_amd import (
    RandomForestClassificationModel,
    RandomForestClassifier,
    RandomForestRegressor,
)
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator
from spark_rapids_ml_amd.parallel.context import get_comm

comm = get_comm()
rng = np.random.default_rng(comm.rank)
n, d = 50_000, 32
X = rng.normal(size=(n, d)).astype(np.float32)
y = (X[:, 0] + 0.5 * X[:, 1] > 0).astype(np.float64)
df = DataFrame({"features": X, "label": y})

rfc = RandomForestClassifier(numTrees=16, maxDepth=8, maxBins=64, seed=7)
model = rfc.fit(df)
out = model.transform(df)
acc = MulticlassClassificationEvaluator(metricName="accuracy").evaluate(out)
if comm.rank == 0:
    print(f"forest: {model.numTrees} trees, {model.totalNumNodes} nodes")
    print(f"train accuracy: {acc:.3f}")
    imp = model.featureImportances
    print("top features:", np.argsort(imp)[::-1][:4].tolist())
    print("single-vector predict:", model.predict(X[0]), "probs:", model.predictProbability(X[0]))

model.write().overwrite().save("/tmp/rfc_model")
loaded = RandomForestClassificationModel.load("/tmp/rfc_model")
assert loaded.numTrees == model.numTrees

# regression variant
yr = (X @ rng.normal(size=d)).astype(np.float64)
reg = RandomForestRegressor(numTrees=8, maxDepth=6, seed=7).fit(
    DataFrame({"features": X, "label": yr})
)
if comm.rank == 0:
    print("regressor trained,", reg.numTrees, "trees")
