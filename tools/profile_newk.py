import sys

import numpy as np

sys.path.insert(0, ".")
from spark_rapids_ml_amd import DBSCAN, UMAP
from spark_rapids_ml_amd.data import DataFrame

rng = np.random.default_rng(0)
n = 200000
centers = rng.normal(size=(20, 64)).astype(np.float32) * 6
X = centers[rng.integers(0, 20, n)] + rng.normal(size=(n, 64)).astype(np.float32)
m = DBSCAN(eps=2.5, min_samples=5).fit(DataFrame.from_numpy(X))
out = m.transform(DataFrame.from_numpy(X))
print("dbscan clusters:", len(set(np.asarray(out["prediction"]).tolist()) - {-1}))
um = UMAP(n_neighbors=15, n_epochs=200, random_state=1).fit(DataFrame.from_numpy(X[:100000]))
print("umap done", um.embedding.shape)
