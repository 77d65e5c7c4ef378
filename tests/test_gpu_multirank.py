"""Multi-rank GBDT on ONE MI355X — the simulated-collective harness.

Two processes share cuda:0 and train the sharded GBDT through the REAL
distributed code path: device binned shards, the C++ arena grower, and the
per-split int64 histogram all_reduce issued from C++ via the c10d
ProcessGroup API (no GIL).  Backend is RCCL when it accepts two ranks on
one device, else gloo (same c10d C++ call sites; RCCL's own enqueue path
is additionally covered by the world_size=1 nccl smoke below).

This is the SURVEY §4 "simulated-collective backend" item: histogram-sync
logic testable without an 8-GPU lease.  The reference's analog is
multi-partition local[*] testing of the socket fabrics
(VerifyLightGBMClassifier.scala executionModeFuncs).
"""
import json
import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs ROCm GPU")


def _make_data(seed, n=60_000, nf=20):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = ((X @ w + rng.normal(size=n) * 0.5) > 0).astype(np.float32)
    return X, y


def _init_pg(backend, rank, world, port):
    torch.distributed.init_process_group(
        backend, init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world)
    if backend == "nccl":
        # force the collective onto the device before training so a
        # duplicate-GPU refusal surfaces here, not mid-train
        t = torch.ones(4, device="cuda")
        torch.distributed.all_reduce(t)
        torch.cuda.synchronize()


def _worker(rank, world, port, backend, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.cuda.set_device(0)  # both ranks share the single GPU
        _init_pg(backend, rank, world, port)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm

        X, y = _make_data(0)
        n_shard = len(X) // world
        sl = slice(rank * n_shard, (rank + 1) * n_shard)
        Xt = torch.from_numpy(X[sl]).cuda()
        yt = torch.from_numpy(y[sl]).cuda()
        comm = Comm()
        cfg = TrainConfig(num_iterations=10, num_leaves=31, seed=7)
        booster, stats = train_booster(Xt, yt, cfg, make_objective("binary"),
                                       comm)
        torch.cuda.synchronize()
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


def _run_world(backend, port, world=2):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, world, port, backend, q))
             for r in range(world)]
    for p in procs:
        p.start()
    results = {}
    try:
        for _ in range(world):
            rank, s = q.get(timeout=420)
            results[rank] = s
    finally:
        for p in procs:
            p.join(timeout=60)
            if p.is_alive():
                p.terminate()
    return results


@requires_gpu
@pytest.mark.timeout(600)
def test_multirank_one_gpu_native_grower_bit_exact():
    """2 ranks on one MI355X: the C++ grower's GIL-free histogram all_reduce
    must give BIT-IDENTICAL boosters on every rank, and match the quality
    of a single-rank run.  Tries RCCL first; falls back to gloo (same c10d
    C++ call sites) when RCCL refuses two ranks on one device."""
    backend_used = "nccl"
    try:
        results = _run_world("nccl", 29951)
        if any(str(s).startswith("ERROR") for s in results.values()):
            raise RuntimeError(str(results))
    except Exception as e:
        print(f"[multirank] nccl 2-ranks-1-GPU unavailable ({e!r}); "
              "falling back to gloo backend with device tensors")
        backend_used = "gloo"
        results = _run_world("gloo", 29952)
    assert not any(str(s).startswith("ERROR") for s in results.values()), results
    print(f"[multirank] backend={backend_used}")

    # bit-exact across ranks: fixed-point integer histograms + exact-sum
    # all_reduce make synchronized growth deterministic
    assert results[0] == results[1]

    d0 = json.loads(results[0])
    assert len(d0["trees"]) == 10

    # quality: the distributed model must match a single-rank full-data model
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.models.gbdt.booster import Booster
    from mmlspark_amd.models.gbdt.objectives import make_objective
    from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
    from mmlspark_amd.parallel.comm import Comm

    X, y = _make_data(0)
    Xt = torch.from_numpy(X).cuda()
    yt = torch.from_numpy(y).cuda()
    single, _ = train_booster(Xt, yt, TrainConfig(num_iterations=10,
                                                  num_leaves=31, seed=7),
                              make_objective("binary"), Comm())
    dist_b = Booster.load_from_string(results[0])
    p_d = torch.sigmoid(dist_b.predict_raw(Xt).squeeze(-1)).cpu().numpy()
    p_s = torch.sigmoid(single.predict_raw(Xt).squeeze(-1)).cpu().numpy()
    auc_d = roc_auc_score(y, p_d)
    auc_s = roc_auc_score(y, p_s)
    print(f"[multirank] auc dist={auc_d:.4f} single={auc_s:.4f}")
    assert auc_d > auc_s - 0.01


def _worker_ws1_rccl(q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = "29953"
        os.environ["RANK"] = "0"
        os.environ["WORLD_SIZE"] = "1"
        torch.cuda.set_device(0)
        torch.distributed.init_process_group("nccl", rank=0, world_size=1)
        from mmlspark_amd.ops import _hip_grower
        pg = torch.distributed.group.WORLD
        t = torch.arange(1024, dtype=torch.int64, device="cuda")
        # world_size==1 → grower skips the reduce; call the C++ path directly
        _hip_grower.allreduce_native(pg, t)
        torch.cuda.synchronize()
        ok = bool(torch.equal(t.cpu(),
                              torch.arange(1024, dtype=torch.int64)))
        q.put("OK" if ok else "MISMATCH")
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put(f"ERROR: {e!r}")


@requires_gpu
@pytest.mark.timeout(300)
def test_rccl_c10d_native_path_smoke():
    """ProcessGroupNCCL(=RCCL) through the grower's C++ allreduce call on a
    device int64 tensor — proves the RCCL enqueue/stream path itself runs
    on MI355X even when 2-ranks-1-GPU is refused."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    p = ctx.Process(target=_worker_ws1_rccl, args=(q,))
    p.start()
    res = q.get(timeout=240)
    p.join(timeout=60)
    assert res == "OK", res
