"""KMeans end-to-end: generate data, fit, evaluate, persist.

Single process:  python examples/kmeans_example.py
Multi-GPU:       python -m torch.distributed.run --nproc-per-node 8 \
                     --master-addr 127.0.0.1 examples/kmeans_example.py
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from spark_rapids_ml_amd import KMeans, KMeansModel
from spark_rapids_ml_amd.data import DataFrame
from spark_rapids_ml_amd.parallel.context import get_comm

comm = get_comm()
rng = np.random.default_rng(comm.rank)
X = rng.normal(size=(100_000, 64)).astype(np.float32)
df = DataFrame.from_numpy(X)

model = KMeans(k=16, maxIter=20, seed=1).fit(df)
out = model.transform(df)
if comm.rank == 0:
    print("centers:", model.cluster_centers_.shape)
    print("training cost:", model.trainingCost)
    print("cluster sizes:", model.summary.clusterSizes[:4], "...")
model.write().overwrite().save("/tmp/km_model")
loaded = KMeansModel.load("/tmp/km_model")
assert np.allclose(loaded.cluster_centers_, model.cluster_centers_)
