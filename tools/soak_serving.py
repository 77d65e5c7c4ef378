"""Serving soak: sustained mixed load against the process cluster with
worker kills + restarts mid-run.  Pass criteria: zero failed requests."""
import json, os, sys, tempfile, threading, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, pandas as pd, requests, torch
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
from mmlspark_amd.serving.server import ProcessServingCluster

DURATION = float(sys.argv[1]) if len(sys.argv) > 1 else 120.0
rng = np.random.default_rng(0)
X = rng.normal(size=(20000, 28)).astype(np.float32)
y = (X[:, 0] > 0).astype(np.float32)
model = LightGBMClassifier(numIterations=100, numLeaves=31,
                           device="cuda" if torch.cuda.is_available() else "cpu"
                           ).fit(pd.DataFrame({"features": list(X), "label": y}))
mdir = tempfile.mkdtemp() + "/m"
model.save(mdir)
cluster = ProcessServingCluster(mdir, n_workers=3, output_cols="prediction",
                                mode="micro-batch", reply_timeout=20).start()
url = f"http://127.0.0.1:{cluster.head.port}/"
stop = threading.Event()
stats = {"ok": 0, "fail": 0}
lock = threading.Lock()
lat = []

def client(ci):
    s = requests.Session()
    while not stop.is_set():
        x = X[rng.integers(0, len(X))].tolist()
        t0 = time.perf_counter()
        try:
            r = s.post(url, json={"features": x}, timeout=25)
            ok = r.status_code == 200 and "prediction" in r.json()
        except Exception:
            ok = False
        with lock:
            stats["ok" if ok else "fail"] += 1
            lat.append(time.perf_counter() - t0)

threads = [threading.Thread(target=client, args=(i,)) for i in range(12)]
t_start = time.perf_counter()
for t in threads:
    t.start()
kills = 0
while time.perf_counter() - t_start < DURATION:
    time.sleep(max(5.0, DURATION / 6))
    if time.perf_counter() - t_start >= DURATION:
        break
    victim = kills % 3
    cluster.kill_worker(victim)     # hard process kill under live load
    time.sleep(2.0)
    cluster.restart_worker(victim)
    kills += 1
stop.set()
for t in threads:
    t.join(timeout=30)
cluster.stop()
lat.sort()
q = lambda p: lat[min(int(p * len(lat)), len(lat) - 1)] * 1e3 if lat else -1
print(json.dumps({
    "soak_seconds": round(time.perf_counter() - t_start, 1),
    "requests_ok": stats["ok"], "requests_failed": stats["fail"],
    "worker_kills": kills, "clients": 12,
    "p50_ms": round(q(0.5), 2), "p99_ms": round(q(0.99), 2),
    "req_per_sec": round(stats["ok"] / (time.perf_counter() - t_start), 1),
}))
assert stats["fail"] == 0, stats
