"""Multi-process distributed GBDT over gloo (world_size=2, CPU).

Covers the RCCL sync logic (histogram all_reduce, shared binning, identical
growth on all ranks) with the gloo backend so it runs without GPUs — the
simulated-collective testing SURVEY §4 calls for.
"""
import json
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp


def _make_data(seed, n=3000, nf=8):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = ((X @ w + rng.normal(size=n) * 0.5) > 0).astype(np.float32)
    return X, y


def _worker_gbdt(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm

        X, y = _make_data(0, n=4000)
        # shard rows across ranks
        sl = slice(rank * 2000, (rank + 1) * 2000)
        Xt = torch.from_numpy(X[sl])
        yt = torch.from_numpy(y[sl])
        comm = Comm()
        cfg = TrainConfig(num_iterations=10, num_leaves=15, seed=7)
        booster, _ = train_booster(Xt, yt, cfg, make_objective("binary"), comm)
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(180)
def test_distributed_gbdt_identical_models_and_quality():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29871
    procs = [ctx.Process(target=_worker_gbdt, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=150)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), results
    # every rank must hold the identical model (synchronized growth)
    t0 = json.loads(results[0])["trees"]
    t1 = json.loads(results[1])["trees"]
    assert len(t0) == len(t1) == 10
    for a, b in zip(t0, t1):
        assert a["feature"] == b["feature"]
        assert a["thr_bin"] == b["thr_bin"]
        np.testing.assert_allclose(a["value"], b["value"], rtol=1e-5, atol=1e-6)

    # quality: distributed model ≈ single-process model on the full data
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.models.gbdt.booster import Booster
    from mmlspark_amd.models.gbdt.objectives import make_objective
    from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
    from mmlspark_amd.parallel.comm import Comm

    X, y = _make_data(0, n=4000)
    dist_booster = Booster.load_from_string(results[0])
    p_dist = torch.sigmoid(dist_booster.predict_raw(torch.from_numpy(X))
                           .squeeze(-1)).numpy()
    single, _ = train_booster(torch.from_numpy(X), torch.from_numpy(y),
                              TrainConfig(num_iterations=10, num_leaves=15,
                                          seed=7),
                              make_objective("binary"), Comm())
    p_single = torch.sigmoid(single.predict_raw(torch.from_numpy(X))
                             .squeeze(-1)).numpy()
    auc_d = roc_auc_score(y, p_dist)
    auc_s = roc_auc_score(y, p_single)
    assert abs(auc_d - auc_s) < 0.02, (auc_d, auc_s)


def _worker_vw(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        import pandas as pd
        from mmlspark_amd.core.schema import SparseVector
        from mmlspark_amd.models.vw.estimators import VowpalWabbitClassifier

        rng = np.random.default_rng(rank)
        size = 1 << 14
        feat_ids = np.random.default_rng(0).integers(0, size, 100)
        w_true = np.zeros(size)
        w_true[feat_ids] = np.random.default_rng(1).normal(size=len(feat_ids))
        rows, labels = [], []
        for _ in range(1500):
            idx = np.unique(rng.choice(feat_ids, size=20))
            val = rng.normal(size=len(idx)).astype(np.float32)
            rows.append(SparseVector(size, idx.astype(np.int32), val))
            labels.append(1.0 if (w_true[idx] * val).sum() > 0 else 0.0)
        df = pd.DataFrame({"features": rows, "label": labels})
        m = VowpalWabbitClassifier(numPasses=3, numBits=14).fit(df)
        q.put((rank, m.weights.tobytes()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(180)
def test_distributed_vw_weights_synchronized():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_vw, args=(r, 2, 29872, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, w = q.get(timeout=150)
        results[rank] = w
    for p in procs:
        p.join(timeout=30)
    assert not any(isinstance(w, str) and w.startswith("ERROR")
                   for w in results.values()), results
    # end-of-pass all_reduce must leave every rank with identical weights
    w0 = np.frombuffer(results[0], dtype=np.float32)
    w1 = np.frombuffer(results[1], dtype=np.float32)
    np.testing.assert_allclose(w0, w1, rtol=1e-6, atol=1e-7)
    assert np.abs(w0).max() > 0


def _worker_ddp_vision(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        import pandas as pd
        from mmlspark_amd.models.image_featurizer import DeepVisionClassifier

        rng = np.random.default_rng(rank)
        imgs, ys = [], []
        for i in range(16):
            img = np.zeros((24, 24, 3), dtype=np.uint8)
            c = i % 2
            img[:, :, 0 if c == 0 else 2] = 200
            imgs.append(img)
            ys.append(c)
        df = pd.DataFrame({"image": imgs, "label": ys})
        torch.manual_seed(0)
        m = DeepVisionClassifier(modelName="ResNet18", imageSize=24, epochs=2,
                                 batchSize=8, device="cpu").fit(df)
        sd = m.module.state_dict()
        q.put((rank, float(sum(v.double().sum() for v in sd.values()))))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(240)
def test_ddp_deep_vision_replicas_agree():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_ddp_vision, args=(r, 2, 29873, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=200)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(isinstance(s, str) for s in results.values()), results
    # DDP gradient sync keeps replicas identical
    assert abs(results[0] - results[1]) < 1e-3, results


def _worker_voting(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm

        X, y = _make_data(0, n=4000, nf=20)
        sl = slice(rank * 2000, (rank + 1) * 2000)
        cfg = TrainConfig(num_iterations=8, num_leaves=15, seed=7,
                          parallelism="voting_parallel", top_k=5)
        booster, _ = train_booster(torch.from_numpy(X[sl]),
                                   torch.from_numpy(y[sl]), cfg,
                                   make_objective("binary"), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(180)
def test_voting_parallel_identical_and_accurate():
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_voting, args=(r, 2, 29874, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=150)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), results
    assert results[0] == results[1]  # ranks agree bit-for-bit
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.models.gbdt.booster import Booster
    X, y = _make_data(0, n=4000, nf=20)
    b = Booster.load_from_string(results[0])
    p = torch.sigmoid(b.predict_raw(torch.from_numpy(X)).squeeze(-1)).numpy()
    assert roc_auc_score(y, p) > 0.9


def _free_port():
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _bfgs_worker(rank, world, port, out):
    import os
    import numpy as np
    import pandas as pd
    import torch.distributed as dist
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from mmlspark_amd.models.vw.estimators import VowpalWabbitClassifier
        rng = np.random.default_rng(100 + rank)  # DIFFERENT shard per rank
        X = rng.normal(size=(1500, 6)).astype(np.float32)
        w = np.array([2, -1, 1, 0, 0, 0], dtype=np.float32)
        y = (X @ w > 0).astype(np.float32)
        df = pd.DataFrame({"features": list(X), "label": y})
        m = VowpalWabbitClassifier(bfgs=True, lossFunction="logistic",
                                   maxIterBfgs=40).fit(df)
        out[rank] = (m.weights.copy(),
                     (m.transform(df)["prediction"].to_numpy() == y).mean())
    finally:
        dist.destroy_process_group()


def test_distributed_bfgs_identical_weights():
    """--bfgs multi-rank: averaged loss + gradient keep every rank's L-BFGS
    trajectory in lockstep — final weight tables identical, model good."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        out = mgr.dict()
        port = _free_port()
        ps = [ctx.Process(target=_bfgs_worker, args=(r, 2, port, out))
              for r in range(2)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(240)
        assert all(p.exitcode == 0 for p in ps), [p.exitcode for p in ps]
        w0, acc0 = out[0]
        w1, acc1 = out[1]
    np.testing.assert_allclose(w0, w1, rtol=0, atol=0)  # bit-identical
    assert acc0 > 0.9 and acc1 > 0.9


@pytest.mark.timeout(240)
def test_distributed_gbdt_ws4_identical():
    """world_size=4 (the scale bench's mid point): synchronized growth still
    yields the identical booster on every rank."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = _free_port()
    procs = [ctx.Process(target=_worker_gbdt_ws4, args=(r, 4, port, q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, s = q.get(timeout=200)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        {k: str(v)[:200] for k, v in results.items()}
    base = results[0]
    for r in range(1, 4):
        assert results[r] == base  # byte-identical model JSON


def _worker_gbdt_ws4(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm

        X, y = _make_data(0, n=4000)
        sl = slice(rank * 1000, (rank + 1) * 1000)
        booster, _ = train_booster(
            torch.from_numpy(X[sl]), torch.from_numpy(y[sl]),
            TrainConfig(num_iterations=6, num_leaves=15, seed=3),
            make_objective("binary"), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


def _worker_native_allreduce(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.ops import _hip_grower
        from mmlspark_amd.parallel.comm import Comm

        comm = Comm()
        pg = comm.native_group()
        assert pg is not None
        # int64 tensor — the histogram dtype the grower reduces
        t = torch.arange(16, dtype=torch.int64) * (rank + 1)
        _hip_grower.allreduce_native(pg, t)
        q.put((rank, t.tolist()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(120)
def test_native_c10d_allreduce_binding():
    """The C++ grower reduces histograms through the c10d ProcessGroup C++
    API (no Python callback, no GIL).  Prove the exact binding it uses —
    pg cast + allreduce + wait on an int64 tensor — at gloo world_size=2."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_native_allreduce,
                         args=(r, 2, 29874, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=100)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), results
    expect = [int(i) * 3 for i in range(16)]  # (rank0 1x + rank1 2x)
    assert results[0] == expect
    assert results[1] == expect


def _worker_vw_ws4(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        import pandas as pd
        from mmlspark_amd.models.vw.estimators import VowpalWabbitRegressor

        rng = np.random.default_rng(100 + rank)
        n, d = 1500, 24
        X = rng.normal(size=(n, d)).astype(np.float32)
        w = np.arange(1, d + 1, dtype=np.float32) / d  # same target all ranks
        y = (X @ w).astype(np.float32)
        df = pd.DataFrame({"features": list(X), "label": y})
        m = VowpalWabbitRegressor(numPasses=4, learningRate=0.3, numBits=10,
                                  adaptive=True, holdoutOff=True).fit(df)
        stats = m.getPerformanceStatistics().iloc[0]
        q.put((rank, {
            "weights_sum": float(np.abs(m.weights).sum()),
            "multipass_ns": int(stats["multipassTimeNs"]),
            "learn_ns": int(stats["learnTimeNs"]),
        }))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(240)
def test_vw_multipass_allreduce_ws4():
    """the per-pass weight all_reduce at world_size=4 —
    every rank converges to the identical averaged table, and the measured
    sync cost is reported via the perf-stats DataFrame (multipassTimeNs)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker_vw_ws4, args=(r, 4, 29877, q))
             for r in range(4)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(4):
        rank, s = q.get(timeout=200)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(isinstance(s, str) for s in results.values()), results
    sums = [results[r]["weights_sum"] for r in range(4)]
    assert max(sums) - min(sums) < 1e-4, sums  # identical synced tables
    costs = [results[r]["multipass_ns"] for r in range(4)]
    assert all(c > 0 for c in costs)
    print("[vw ws4] per-pass allreduce cost ns:", costs)


def _worker_gbdt_cat(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm

        rng = np.random.default_rng(0)
        n = 4000
        Xnum = rng.normal(size=(n, 4)).astype(np.float32)
        cat = rng.integers(0, 12, size=n)
        # non-monotone categorical effect forces one-vs-rest set splits
        eff = np.array([1.5, -2, 0.3, 2, -1, 0, 1, -1.5, 0.7, -0.4, 2.2, -2.5])
        y = ((Xnum[:, 0] + eff[cat] + rng.normal(size=n) * 0.4) > 0) \
            .astype(np.float32)
        X = np.concatenate([cat[:, None].astype(np.float32), Xnum], axis=1)
        sl = slice(rank * 2000, (rank + 1) * 2000)
        cfg = TrainConfig(num_iterations=8, num_leaves=15, seed=7,
                          categorical_features=[0])
        booster, _ = train_booster(torch.from_numpy(X[sl]),
                                   torch.from_numpy(y[sl]), cfg,
                                   make_objective("binary"), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(180)
def test_distributed_categorical_identical_models():
    """Categorical one-vs-rest splits under ws=2 histogram sync: every rank
    must grow the identical tree sequence (cat scan is deterministic over
    the REDUCED histogram, so bitsets agree bit-for-bit)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29911
    procs = [ctx.Process(target=_worker_gbdt_cat, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=150)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    assert results[0] == results[1]  # byte-identical boosters
    t = json.loads(results[0])["trees"]
    assert len(t) == 8
    # at least one categorical (set) split must actually appear:
    # cat_offset[node] >= 0 marks a 256-bit bitset split
    assert any(any(o >= 0 for o in tree.get("cat_offset", []))
               for tree in t), "no categorical split in any tree"


def _worker_gbdt_boosting(rank, world, port, q, boosting):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm
        X, y = _make_data(0, n=4000)
        sl = slice(rank * 2000, (rank + 1) * 2000)
        cfg = TrainConfig(num_iterations=8, num_leaves=15, seed=7,
                          boosting=boosting,
                          bagging_fraction=0.7 if boosting == "rf" else 1.0,
                          bagging_freq=1 if boosting == "rf" else 0)
        booster, _ = train_booster(torch.from_numpy(X[sl]),
                                   torch.from_numpy(y[sl]), cfg,
                                   make_objective("binary"), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(300)
@pytest.mark.parametrize("boosting", ["goss", "dart", "rf"])
def test_distributed_boosting_variants_identical(boosting):
    """goss/dart/rf under ws=2 histogram sync: per-rank row sampling uses
    rank-seeded RNGs (data-parallel), yet reduced histograms must drive
    every rank to the byte-identical booster."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29920 + {"goss": 0, "dart": 1, "rf": 2}[boosting]
    procs = [ctx.Process(target=_worker_gbdt_boosting,
                         args=(r, 2, port, q, boosting)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=250)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    assert results[0] == results[1]
    assert len(json.loads(results[0])["trees"]) >= 8


def _worker_ranker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        import pandas as pd
        from mmlspark_amd.models.gbdt.estimators import LightGBMRanker
        rng = np.random.default_rng(0)
        nq, per_q = 200, 10
        feats = rng.normal(size=(nq * per_q, 6)).astype(np.float32)
        rel = feats[:, 0] + 0.3 * rng.normal(size=nq * per_q)
        labels = np.digitize(rel, np.quantile(rel, [0.5, 0.8, 0.95])) \
            .astype(np.float64)
        groups = np.repeat(np.arange(nq), per_q)
        # whole query groups per rank (repartitionByGroupingColumn shape)
        mask = (groups % world) == rank
        df = pd.DataFrame({"group": groups[mask],
                           "features": list(feats[mask]),
                           "label": labels[mask]})
        m = LightGBMRanker(numIterations=6, numLeaves=15,
                           groupCol="group").fit(df)
        q.put((rank, m.booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(300)
def test_distributed_ranker_identical_models():
    """LambdaRank under ws=2: per-rank NDCG gradients over local query
    groups, reduced histograms → byte-identical boosters on both ranks."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29931
    procs = [ctx.Process(target=_worker_ranker, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=250)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    assert results[0] == results[1]


def _worker_earlystop(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm
        X, y = _make_data(0, n=4000)
        Xv, yv = _make_data(99, n=800)  # same valid set on every rank
        sl = slice(rank * 2000, (rank + 1) * 2000)
        cfg = TrainConfig(num_iterations=60, num_leaves=31, seed=7,
                          early_stopping_round=5)
        booster, _ = train_booster(
            torch.from_numpy(X[sl]), torch.from_numpy(y[sl]), cfg,
            make_objective("binary"), Comm(),
            valid_sets=[(torch.from_numpy(Xv), torch.from_numpy(yv), None)])
        q.put((rank, json.dumps({"best": booster.best_iteration,
                                 "model": booster.save_to_string()})))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(300)
def test_distributed_early_stopping_agrees():
    """Early stopping under ws=2: the shared validation set gives every
    rank the same metric stream, so best_iteration and the final booster
    must agree exactly (a rank divergence here would deadlock real runs)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29941
    procs = [ctx.Process(target=_worker_earlystop, args=(r, 2, port, q))
             for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=250)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    a, b = json.loads(results[0]), json.loads(results[1])
    assert a["best"] == b["best"]
    assert a["model"] == b["model"]
    assert a["best"] is not None and a["best"] < 60  # it actually stopped


def _worker_uneven(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm
        X, y = _make_data(0, n=4000)
        bounds = [0, 1800, 3300, 4000]  # deliberately unequal shards
        sl = slice(bounds[rank], bounds[rank + 1])
        cfg = TrainConfig(num_iterations=6, num_leaves=15, seed=7)
        booster, _ = train_booster(torch.from_numpy(X[sl]),
                                   torch.from_numpy(y[sl]), cfg,
                                   make_objective("binary"), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(300)
def test_distributed_uneven_shards_ws3():
    """World size 3 with unequal shard sizes (1800/1500/700): reduced
    histograms weight rows correctly regardless of shard balance and all
    ranks grow the identical booster — the empty/uneven-partition
    robustness of LightGBMBase.scala:346-354 in distributed form."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29951
    procs = [ctx.Process(target=_worker_uneven, args=(r, 3, port, q))
             for r in range(3)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(3):
        rank, s = q.get(timeout=250)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    assert results[0] == results[1] == results[2]


def _worker_objective(rank, world, port, q, objective):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm
        rng = np.random.default_rng(0)
        X = rng.normal(size=(3000, 6)).astype(np.float32)
        y = np.exp(X[:, 0] * 0.5 + rng.normal(size=3000) * 0.2) \
            .astype(np.float32)  # positive labels for poisson/tweedie
        # deliberately skewed shards so shard-local label stats DIFFER
        sl = slice(0, 2000) if rank == 0 else slice(2000, 3000)
        cfg = TrainConfig(num_iterations=4, num_leaves=15, seed=7)
        booster, _ = train_booster(torch.from_numpy(X[sl]),
                                   torch.from_numpy(y[sl]), cfg,
                                   make_objective(objective), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(300)
@pytest.mark.parametrize("objective", ["poisson", "tweedie", "regression_l1"])
def test_distributed_init_score_consistency(objective):
    """Objectives whose init score depends on label statistics (log-mean,
    median): skewed shards give different LOCAL stats, yet the stored
    base_score — and therefore the whole serialized model — must agree."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29971 + {"poisson": 0, "tweedie": 1, "regression_l1": 2}[objective]
    procs = [ctx.Process(target=_worker_objective,
                         args=(r, 2, port, q, objective)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=250)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    assert results[0] == results[1]
