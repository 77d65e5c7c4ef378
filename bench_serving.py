#!/usr/bin/env python3
"""Serving p50 latency benchmark (BASELINE config #6: reference bar is
'sub-millisecond' continuous-mode latency, docs/mmlspark-serving.md:10).

Trains a small GBDT, serves it in continuous mode through the
hipGraph-captured low-latency scorer, and measures end-to-end HTTP p50/p90/p99
over loopback plus the raw scorer latency (no HTTP)."""
import argparse
import json
import time

import numpy as np
import pandas as pd


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--requests", type=int, default=2000)
    ap.add_argument("--trees", type=int, default=100)
    ap.add_argument("--features", type=int, default=28)
    ap.add_argument("--clients", type=int, default=0,
                    help="also run a concurrent micro-batch phase with N "
                         "keep-alive clients (VERDICT r1: 64+, report p99)")
    args = ap.parse_args()

    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.serving.server import LowLatencyGBDTScorer, ServingServer

    rng = np.random.default_rng(0)
    n, nf = 20000, args.features
    X = rng.normal(size=(n, nf)).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 0).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    model = LightGBMClassifier(numIterations=args.trees, numLeaves=31).fit(df)

    scorer = LowLatencyGBDTScorer(model.booster, max_batch=1)
    x0 = X[0]

    # raw scorer latency (no HTTP)
    for _ in range(50):
        scorer.score(x0[None, :])
    lat_raw = []
    for _ in range(args.requests):
        t0 = time.perf_counter()
        scorer.score(x0[None, :])
        lat_raw.append((time.perf_counter() - t0) * 1e3)

    srv = ServingServer(scorer, port=0, mode="continuous").start()
    try:
        import http.client
        import socket
        conn = http.client.HTTPConnection("127.0.0.1", srv.port)
        conn.connect()
        conn.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        body = json.dumps({"features": x0.tolist()})
        hdrs = {"Content-Type": "application/json"}

        def one():
            conn.request("POST", "/", body=body, headers=hdrs)
            r = conn.getresponse()
            r.read()
            return r.status

        for _ in range(100):
            assert one() == 200
        lat = []
        for _ in range(args.requests):
            t0 = time.perf_counter()
            code = one()
            lat.append((time.perf_counter() - t0) * 1e3)
            assert code == 200
    finally:
        srv.stop()

    lat.sort()
    lat_raw.sort()
    q = lambda a, p: a[int(p * len(a))]
    print(json.dumps({
        "metric": "serving_p50_latency_ms",
        "value": q(lat, 0.5),
        "unit": "ms",
        "higher_is_better": False,
        "http_p50_ms": q(lat, 0.5),
        "http_p90_ms": q(lat, 0.9),
        "http_p99_ms": q(lat, 0.99),
        "scorer_p50_ms": q(lat_raw, 0.5),
        "scorer_p99_ms": q(lat_raw, 0.99),
        "hipgraph": scorer.graph is not None,
        "config": {"trees": args.trees, "features": args.features,
                   "mode": "continuous", "transport": "loopback HTTP"},
        "reference_bar": "sub-millisecond (docs/mmlspark-serving.md:10)",
    }), flush=True)

    if args.clients > 0:
        concurrent_phase(model, X, args)


def concurrent_phase(model, X, args):
    """Micro-batch under N concurrent keep-alive clients: req/s + p50/p99."""
    import http.client
    import socket
    import threading

    from mmlspark_amd.serving.server import LowLatencyGBDTScorer, ServingServer

    scorer = LowLatencyGBDTScorer(model.booster,
                                  max_batch=max(64, args.clients))

    def handler(payloads):
        Xb = np.asarray([p["features"] for p in payloads], dtype=np.float32)
        scores = scorer.score(Xb)
        return [{"score": s.tolist()} for s in scores]

    srv = ServingServer(handler, port=0, mode="micro-batch",
                        max_batch=max(64, args.clients),
                        batch_wait_ms=0.5).start()
    n_clients = args.clients
    per_client = max(50, args.requests // n_clients)
    lats = [[] for _ in range(n_clients)]
    errs = [0] * n_clients
    barrier = threading.Barrier(n_clients + 1)

    def client(ci):
        conn = http.client.HTTPConnection("127.0.0.1", srv.port)
        conn.connect()
        conn.sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        body = json.dumps({"features": X[ci % len(X)].tolist()})
        hdrs = {"Content-Type": "application/json"}
        for _ in range(10):  # warm
            conn.request("POST", "/", body=body, headers=hdrs)
            conn.getresponse().read()
        barrier.wait()
        for _ in range(per_client):
            t0 = time.perf_counter()
            conn.request("POST", "/", body=body, headers=hdrs)
            r = conn.getresponse()
            r.read()
            lats[ci].append((time.perf_counter() - t0) * 1e3)
            if r.status != 200:
                errs[ci] += 1
        conn.close()

    import threading as _th
    threads = [_th.Thread(target=client, args=(i,))
               for i in range(n_clients)]
    for t in threads:
        t.start()
    barrier.wait()
    t0 = time.perf_counter()
    for t in threads:
        t.join()
    wall = time.perf_counter() - t0
    srv.stop()
    lat = sorted(x for l in lats for x in l)
    q = lambda a, p: a[min(int(p * len(a)), len(a) - 1)]
    print(json.dumps({
        "metric": "serving_concurrent_req_per_sec",
        "value": n_clients * per_client / wall,
        "unit": "req/s",
        "higher_is_better": True,
        "clients": n_clients,
        "requests": n_clients * per_client,
        "errors": sum(errs),
        "p50_ms": q(lat, 0.5),
        "p90_ms": q(lat, 0.9),
        "p99_ms": q(lat, 0.99),
        "hipgraph": scorer.graph is not None,
        "config": {"trees": args.trees, "features": args.features,
                   "mode": "micro-batch", "transport": "loopback HTTP"},
    }), flush=True)


if __name__ == "__main__":
    main()
