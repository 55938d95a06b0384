"""LogisticRegression with CV model selection (reference-style workflow)."""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from spark_rapids_ml_amd import LogisticRegression
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator
from spark_rapids_ml_amd.tuning import CrossValidator, ParamGridBuilder

rng = np.random.default_rng(0)
X = rng.normal(size=(50_000, 128))
w = rng.normal(size=128)
y = (X @ w + 0.5 * rng.normal(size=50_000) > 0).astype(np.float64)
df = DataFrame.from_numpy(X.astype(np.float32), y)

lr = LogisticRegression(maxIter=100)
grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [1e-5, 1e-3, 1e-1]).build()
cv = CrossValidator(estimator=lr, estimatorParamMaps=grid,
                    evaluator=MulticlassClassificationEvaluator(metricName="accuracy"),
                    numFolds=3)
model = cv.fit(df)
print("avg accuracy per regParam:", [round(m, 4) for m in model.avgMetrics])
print("best regParam:", model.bestModel.getOrDefault("regParam"))
print("objectiveHistory:", model.bestModel.objectiveHistory[:5], "...")
