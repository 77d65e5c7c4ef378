"""Round-2 features tour: sparse CSR training, Arrow interop, iteration
checkpoints + elastic restart, and process-based serving workers."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import tempfile

import numpy as np
import pandas as pd

from mmlspark_amd.core.interop import arrow_to_pandas, pandas_to_arrow
from mmlspark_amd.core.schema import SparseVector
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier

rng = np.random.default_rng(0)

# --- 1. sparse CSR training (no densify: 5 B/nnz instead of nf B/row) ----
n, nf, nnz = 4000, 5000, 12
rows = []
w = rng.normal(size=nf)
y = np.zeros(n, dtype=np.float32)
for i in range(n):
    idx = np.sort(rng.choice(nf, size=nnz, replace=False)).astype(np.int32)
    val = rng.normal(size=nnz).astype(np.float32)
    y[i] = 1.0 if (w[idx] * val).sum() > 0 else 0.0
    rows.append(SparseVector(nf, idx, val))
df = pd.DataFrame({"features": rows, "label": y})

ckdir = tempfile.mkdtemp()
model = LightGBMClassifier(numIterations=20, numLeaves=15, maxBin=31,
                           minDataInLeaf=5,
                           checkpointDir=ckdir,          # elastic restart
                           checkpointInterval=5).fit(df)  # resumes from here
acc = (model.transform(df)["prediction"].to_numpy() == y).mean()
print(f"sparse {n}x{nf} (nnz={nnz}): train acc {acc:.3f}, "
      f"checkpoint at {ckdir}/checkpoint.json")

# --- 2. Arrow in → Arrow out --------------------------------------------
dense = pd.DataFrame({
    "features": list(rng.normal(size=(1000, 10)).astype(np.float32)),
})
dense["label"] = (np.stack(dense["features"])[:, 0] > 0).astype(np.float64)
table = pandas_to_arrow(dense)
m2 = LightGBMClassifier(numIterations=5, numLeaves=7).fit(table)
scored = m2.transform(table)                    # a pyarrow Table
print("arrow round trip:", type(scored).__name__, scored.column_names[:4])
assert "prediction" in scored.column_names
_ = arrow_to_pandas(scored)

# --- 3. process-based serving cluster with failover ----------------------
import requests

from mmlspark_amd.serving.server import ProcessServingCluster

mdir = tempfile.mkdtemp() + "/model"
m2.save(mdir)
cluster = ProcessServingCluster(mdir, n_workers=2,
                                output_cols="prediction",
                                mode="micro-batch").start()
try:
    url = f"http://127.0.0.1:{cluster.head.port}/"
    x = dense["features"].iloc[0].tolist()
    r = requests.post(url, json={"features": x}, timeout=20)
    print("served:", r.json())
    cluster.kill_worker(0)                      # hard crash one process
    r = requests.post(url, json={"features": x}, timeout=20)
    print("after worker kill (failover):", r.status_code, r.json())
    assert r.status_code == 200
finally:
    cluster.stop()
print("done")
