"""CrossValidator + Pipeline: single-pass multi-model hyper-parameter sweep
over one data load per fold, with the VectorAssembler bypass.

Single process:  python examples/cross_validation_example.py
Multi-GPU:       srml-amd-launch examples/cross_validation_example.py
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from spark_rapids_ml_amd import LogisticRegression
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.evaluation import MulticlassClassificationEvaluator
from spark_rapids_ml_amd.parallel.context import get_comm
from spark_rapids_ml_amd.pipeline import Pipeline, VectorAssembler
from spark_rapids_ml_amd.tuning import CrossValidator, ParamGridBuilder

comm = get_comm()
rng = np.random.default_rng(comm.rank)
n = 20_000
a = rng.normal(size=n).astype(np.float32)
b = rng.normal(size=n).astype(np.float32)
label = (a + 0.5 * b > 0).astype(np.float64)
df = DataFrame({"a": a, "b": b, "label": label})

# Pipeline: the assembler is bypassed; scalar columns feed the GPU estimator
pipe = Pipeline([
    VectorAssembler(["a", "b"]),
    LogisticRegression(maxIter=40),
])
pmodel = pipe.fit(df)
if comm.rank == 0:
    print("pipeline accuracy:",
          (np.asarray(pmodel.transform(df)["prediction"]) == label).mean())

# CrossValidator: all regParam values trained in ONE data pass per fold
lr = LogisticRegression(maxIter=40)
grid = ParamGridBuilder().addGrid(lr.getParam("regParam"), [0.0, 0.001, 0.1]).build()
cv = CrossValidator(lr, grid, MulticlassClassificationEvaluator(metricName="accuracy"), numFolds=3)
cvm = cv.fit(DataFrame({"features": np.column_stack([a, b]), "label": label}))
if comm.rank == 0:
    print("avg metrics per param map:", [round(m, 4) for m in cvm.avgMetrics])
    print("best regParam:", cvm.bestModel.getRegParam())
