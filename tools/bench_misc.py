"""Secondary model-family throughput on MI355X: KNN (MFMA GEMM brute
force), IsolationForest scoring, SAR recommendation, serving concurrency.
Writes one JSON line per benchmark; run on a GPU box, summaries go to
profiles/.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd
import torch


def timeit(fn, warmup=2, iters=5):
    for _ in range(warmup):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def bench_knn():
    from mmlspark_amd.models.knn import KNN
    rng = np.random.default_rng(0)
    n_index, n_query, dim, k = 200_000, 50_000, 128, 10
    idx_df = pd.DataFrame({"features": list(rng.normal(
        size=(n_index, dim)).astype(np.float32))})
    q = pd.DataFrame({"features": list(rng.normal(
        size=(n_query, dim)).astype(np.float32))})
    model = KNN(k=k).fit(idx_df)
    dt = timeit(lambda: model.transform(q))
    print(json.dumps({"bench": "knn_brute_mfma", "queries_per_sec":
                      n_query / dt, "n_index": n_index, "dim": dim, "k": k}))


def bench_conditional_knn():
    """conditional (label-filtered) search at a corpus
    size where the reference's broadcast ball tree would not fit a UDF —
    2M x d128, 16 labels, device bitmask filtering."""
    from mmlspark_amd.models.knn import ConditionalKNN
    rng = np.random.default_rng(1)
    n_index, n_query, dim, k = 2_000_000, 20_000, 128, 10
    idx_df = pd.DataFrame({
        "features": list(rng.normal(size=(n_index, dim)).astype(np.float32)),
        "labels": rng.integers(0, 16, size=n_index),
    })
    q = pd.DataFrame({
        "features": list(rng.normal(size=(n_query, dim)).astype(np.float32)),
        "conditioner": [list(rng.choice(16, size=3, replace=False))
                        for _ in range(n_query)],
    })
    model = ConditionalKNN(k=k, labelCol="labels").fit(idx_df)
    dt = timeit(lambda: model.transform(q), warmup=1, iters=3)
    print(json.dumps({"bench": "conditional_knn_bitmask",
                      "queries_per_sec": n_query / dt,
                      "n_index": n_index, "dim": dim, "k": k,
                      "n_labels": 16}))


def bench_iforest():
    from mmlspark_amd.models.iforest import IsolationForest
    rng = np.random.default_rng(1)
    n, nf = 2_000_000, 16
    train = pd.DataFrame({"features": list(rng.normal(
        size=(20_000, nf)).astype(np.float32))})
    model = IsolationForest(numEstimators=100).fit(train)
    X = pd.DataFrame({"features": list(rng.normal(
        size=(n, nf)).astype(np.float32))})
    dt = timeit(lambda: model.transform(X), warmup=1, iters=3)
    print(json.dumps({"bench": "iforest_score", "rows_per_sec": n / dt,
                      "trees": 100, "features": nf}))


def bench_sar():
    from mmlspark_amd.models.sar import SAR
    rng = np.random.default_rng(2)
    n_users, n_items, n_inter = 50_000, 5_000, 1_000_000
    df = pd.DataFrame({
        "user": rng.integers(0, n_users, n_inter),
        "item": rng.integers(0, n_items, n_inter),
        "rating": rng.random(n_inter).astype(np.float32) * 5,
        "timestamp": rng.integers(1_600_000_000, 1_700_000_000, n_inter),
    })
    model = SAR(userCol="user", itemCol="item", ratingCol="rating",
                timeCol="timestamp").fit(df)
    dt = timeit(lambda: model.recommendForAllUsers(10), warmup=1, iters=3)
    print(json.dumps({"bench": "sar_recommend_all", "users_per_sec":
                      n_users / dt, "items": n_items,
                      "interactions": n_inter}))


def bench_images():
    """ImageTransformer stage-list pipeline (resize/crop/normalize) — the
    opencv-module analog — images/s on device tensors."""
    from mmlspark_amd.models.images import ImageTransformer
    rng = np.random.default_rng(5)
    n = 4096
    imgs = [rng.integers(0, 255, size=(256, 256, 3)).astype(np.uint8)
            for _ in range(n)]
    df = pd.DataFrame({"image": imgs})
    t = (ImageTransformer(inputCol="image", outputCol="out")
         .resize(224, 224).crop(12, 12, 200, 200)
         .normalize(mean=[0.485, 0.456, 0.406], std=[0.229, 0.224, 0.225]))
    dt = timeit(lambda: t.transform(df), warmup=1, iters=3)
    print(json.dumps({"bench": "image_transformer", "images_per_sec": n / dt,
                      "pipeline": "resize224+crop200+normalize"}))


def bench_serving_concurrent():
    import http.client
    import socket
    import threading
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.serving.server import (LowLatencyGBDTScorer,
                                             ServingServer)
    rng = np.random.default_rng(3)
    X = rng.normal(size=(5000, 28)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    m = LightGBMClassifier(numIterations=100, numLeaves=31,
                           device="cuda" if torch.cuda.is_available()
                           else "cpu").fit(
        pd.DataFrame({"features": list(X), "label": y}))
    scorer = LowLatencyGBDTScorer(m.booster, max_batch=64)

    def handler(payloads):  # micro-batch: one kernel launch per epoch batch
        Xb = np.stack([np.asarray(p["features"], dtype=np.float32)
                       for p in payloads])
        out = scorer.score(Xb)
        return [{"p": float(np.asarray(out).reshape(len(payloads), -1)[i, 0])}
                for i in range(len(payloads))]

    srv = ServingServer(handler, port=0, mode="micro-batch",
                        max_batch=64, batch_wait_ms=0.5).start()
    body = json.dumps({"features": X[0].tolist()}).encode()
    n_threads, per = 8, 500

    def client(res, i):
        conn = http.client.HTTPConnection("127.0.0.1", srv.port)
        conn.connect()
        conn.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        t0 = time.perf_counter()
        for _ in range(per):
            conn.request("POST", "/", body,
                         {"Content-Type": "application/json"})
            conn.getresponse().read()
        res[i] = time.perf_counter() - t0
        conn.close()

    res = [0.0] * n_threads
    ts = [threading.Thread(target=client, args=(res, i))
          for i in range(n_threads)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    total = n_threads * per
    print(json.dumps({"bench": "serving_concurrent", "requests_per_sec":
                      total / max(res), "concurrency": n_threads,
                      "model_trees": 100}))
    srv.stop()


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    for name, fn in [("knn", bench_knn),
                     ("cknn", bench_conditional_knn),
                     ("iforest", bench_iforest),
                     ("images", bench_images),
                     ("sar", bench_sar),
                     ("serving", bench_serving_concurrent)]:
        if which in ("all", name):
            fn()
