"""Low-latency model serving: train, hipGraph-capture the scorer, serve over
HTTP, fire a few requests (continuous mode — sub-millisecond p50 on MI355X)."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import numpy as np
import pandas as pd
import requests

from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
from mmlspark_amd.serving.server import LowLatencyGBDTScorer, ServingServer

rng = np.random.default_rng(0)
X = rng.normal(size=(20_000, 28)).astype(np.float32)
y = (X[:, 0] + X[:, 1] > 0).astype(np.float32)
model = LightGBMClassifier(numIterations=100, numLeaves=31).fit(
    pd.DataFrame({"features": list(X), "label": y}))

scorer = LowLatencyGBDTScorer(model.booster, max_batch=1)
srv = ServingServer(scorer, port=0, mode="continuous").start()
url = f"http://127.0.0.1:{srv.port}/"
s = requests.Session()
lat = []
for i in range(200):
    t0 = time.perf_counter()
    r = s.post(url, json={"features": X[i].tolist()})
    lat.append((time.perf_counter() - t0) * 1e3)
lat.sort()
print(f"p50 {lat[100]:.3f} ms   p99 {lat[197]:.3f} ms")
print("sample reply:", r.json())
srv.stop()
