"""Cross-host worker registration: a worker launched independently (another
process, any host) POSTs {"__register__": ServiceInfo} to the cluster head
and joins the round-robin rotation — the DriverServiceUtils rendezvous
(HTTPSourceV2.scala:133-198) closed as a live path (was a round-3 note)."""
import json
import os
import subprocess
import sys

import numpy as np
import pandas as pd
import pytest
import requests


@pytest.fixture()
def model_dir(tmp_path):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(0)
    X = rng.normal(size=(500, 6)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = LightGBMClassifier(numIterations=4, numLeaves=7).fit(df)
    p = str(tmp_path / "m")
    m.save(p)
    return p


def test_remote_worker_registers_and_serves(model_dir):
    from mmlspark_amd.serving.server import ProcessServingCluster
    cluster = ProcessServingCluster(model_dir, n_workers=1,
                                    mode="micro-batch").start()
    try:
        head = f"http://127.0.0.1:{cluster.head.port}/"
        # launch an INDEPENDENT worker (not via the cluster) that
        # self-registers with the head
        proc = subprocess.Popen(
            [sys.executable, "-m", "mmlspark_amd.serving.worker",
             "--model", model_dir, "--port", "0", "--mode", "micro-batch",
             "--name", "offbox-worker", "--report-to", head],
            stdout=subprocess.PIPE, text=True, env=dict(os.environ))
        try:
            ready = json.loads(proc.stdout.readline())
            assert ready["ready"]
            # wait for the registration to land
            for _ in range(50):
                if len(cluster.workers) == 2:
                    break
                import time
                time.sleep(0.1)
            assert len(cluster.workers) == 2
            assert cluster.workers[1].name == "offbox-worker"
            # scoring through the head reaches both workers (round-robin)
            row = {"features": [0.5, 0, 0, 0, 0, 0]}
            for _ in range(4):
                r = requests.post(head, json=row, timeout=10)
                assert r.status_code == 200 and "prediction" in r.json()
            # kill the LOCAL worker: failover must route to the remote one
            cluster.kill_worker(0)
            r = requests.post(head, json=row, timeout=10)
            assert r.status_code == 200
            # registration is idempotent
            info = {"host": "127.0.0.1", "port": ready["port"],
                    "name": "offbox-worker"}
            r = requests.post(head, json={"__register__": info}, timeout=10)
            assert r.json()["known"] is True
            assert len(cluster.workers) == 2
        finally:
            proc.terminate()
            proc.wait(timeout=10)
    finally:
        cluster.stop()
