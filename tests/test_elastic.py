"""Elastic training: iteration-level checkpoints + gang restart (the
training-job fault-tolerance analog of Spark barrier task retry; VERDICT
r1 aux gap "no training-job elastic/retry analog")."""
import json
import os
import subprocess
import sys
import textwrap

import numpy as np
import pandas as pd
import pytest
import torch

from mmlspark_amd.models.gbdt.booster import Booster
from mmlspark_amd.models.gbdt.objectives import make_objective
from mmlspark_amd.models.gbdt.trainer import (TrainConfig, load_checkpoint,
                                              train_booster)
from mmlspark_amd.parallel.comm import Comm


def _data(seed=0, n=3000, nf=10):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = ((X @ w + 0.3 * rng.normal(size=n)) > 0).astype(np.float32)
    return torch.from_numpy(X), torch.from_numpy(y)


def test_checkpoint_write_and_resume(tmp_path):
    X, y = _data()
    cfg = TrainConfig(num_iterations=10, num_leaves=15, seed=1)
    ckdir = str(tmp_path / "ck")

    # full run with checkpoints
    b_full, _ = train_booster(X, y, cfg, make_objective("binary"), Comm(),
                              checkpoint_dir=ckdir, checkpoint_every=3)
    assert b_full.num_trees == 10
    ck = load_checkpoint(ckdir)
    assert ck is not None
    b_ck, it = ck
    assert it == 9 and b_ck.num_trees == 9  # last multiple of 3

    # resume completes only the remaining iteration
    b_res, stats = train_booster(X, y, cfg, make_objective("binary"), Comm(),
                                 checkpoint_dir=ckdir, checkpoint_every=3)
    assert b_res.num_trees == 10
    assert stats.iterations == 1  # only iteration 10 ran


def test_elastic_gang_restart_resumes(tmp_path):
    """Kill training mid-run (fault injection at iteration 5), relaunch via
    run_elastic: the restarted run resumes from the checkpoint and the final
    model matches an uninterrupted run's quality."""
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.utils.elastic import run_elastic

    ckdir = str(tmp_path / "ck")
    out = str(tmp_path / "model.txt")
    script = tmp_path / "train_job.py"
    script.write_text(textwrap.dedent(f"""
        import os, sys
        sys.path.insert(0, {os.path.dirname(os.path.dirname(os.path.abspath(__file__)))!r})
        import numpy as np, torch
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import (TrainConfig,
                                                      TrainingSession,
                                                      load_checkpoint,
                                                      _save_checkpoint)
        from mmlspark_amd.parallel.comm import Comm
        rng = np.random.default_rng(0)
        X = rng.normal(size=(3000, 10)).astype(np.float32)
        w = rng.normal(size=10)
        y = ((X @ w + 0.3 * rng.normal(size=3000)) > 0).astype(np.float32)
        Xt, yt = torch.from_numpy(X), torch.from_numpy(y)
        cfg = TrainConfig(num_iterations=12, num_leaves=15, seed=1)
        ck = load_checkpoint({ckdir!r})
        init, start = (ck if ck else (None, 0))
        ses = TrainingSession(Xt, yt, cfg, make_objective("binary"), Comm(),
                              init_booster=init)
        for it in range(start, cfg.num_iterations):
            ses.step()
            _save_checkpoint({ckdir!r}, ses.booster(), it + 1)
            if it == 5 and not os.environ.get("NO_FAULT"):
                os._exit(17)  # simulated crash mid-training
        with open({out!r}, "w") as f:
            f.write(ses.booster().save_to_string())
    """))

    env = dict(os.environ)
    restarts = run_elastic([sys.executable, str(script)], max_restarts=2,
                           env=env, backoff_s=0.1)
    assert restarts == 1  # crashed once, second attempt resumed + finished
    with open(out) as f:
        b = Booster.load_from_string(f.read())
    assert b.num_trees == 12

    X, y = _data()
    p = torch.sigmoid(b.predict_raw(X).squeeze(-1)).numpy()
    auc_elastic = roc_auc_score(y.numpy(), p)

    # uninterrupted baseline
    b0, _ = train_booster(X, y, TrainConfig(num_iterations=12, num_leaves=15,
                                            seed=1),
                          make_objective("binary"), Comm())
    p0 = torch.sigmoid(b0.predict_raw(X).squeeze(-1)).numpy()
    auc_base = roc_auc_score(y.numpy(), p0)
    assert auc_elastic > auc_base - 0.01, (auc_elastic, auc_base)


def test_estimator_checkpoint_params(tmp_path):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    X, y = _data(seed=3)
    df = pd.DataFrame({"features": list(X.numpy()), "label": y.numpy()})
    ckdir = str(tmp_path / "ck")
    m = LightGBMClassifier(numIterations=6, numLeaves=7,
                           checkpointDir=ckdir,
                           checkpointInterval=2).fit(df)
    assert m.booster.num_trees == 6
    with open(os.path.join(ckdir, "checkpoint.json")) as f:
        d = json.load(f)
    assert d["iteration"] == 6


def test_checkpoints_with_num_batches(tmp_path):
    """numBatches + checkpointDir: each batch checkpoints in its own
    namespace so later batches never 'resume' a finished earlier batch."""
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    X, y = _data(seed=9)
    df = pd.DataFrame({"features": list(X.numpy()), "label": y.numpy()})
    ckdir = str(tmp_path / "ck")
    m = LightGBMClassifier(numIterations=4, numLeaves=7, numBatches=2,
                           checkpointDir=ckdir,
                           checkpointInterval=2).fit(df)
    assert m.booster.num_trees == 8  # 4 per batch — batch 2 actually ran
    assert os.path.exists(os.path.join(ckdir, "batch0", "checkpoint.json"))
    assert os.path.exists(os.path.join(ckdir, "batch1", "checkpoint.json"))


def _dist_ck_worker(rank, world, port, ckdir, n_iters, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm
        rng = np.random.default_rng(0)
        X = rng.normal(size=(4000, 10)).astype(np.float32)
        w = rng.normal(size=10)
        y = ((X @ w) > 0).astype(np.float32)
        sl = slice(rank * 2000, (rank + 1) * 2000)
        cfg = TrainConfig(num_iterations=n_iters, num_leaves=15, seed=2)
        b, _ = train_booster(torch.from_numpy(X[sl]), torch.from_numpy(y[sl]),
                             cfg, make_objective("binary"), Comm(),
                             checkpoint_dir=ckdir, checkpoint_every=2)
        q.put((rank, b.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(240)
def test_distributed_checkpoint_resume_ws2(tmp_path):
    """Gang restart at world_size=2: run to iteration 4 (checkpointing),
    relaunch the gang with num_iterations=8 — every rank resumes from the
    shared checkpoint, stays bit-identical across ranks, and ends with 8
    trees."""
    import torch.multiprocessing as mp
    ckdir = str(tmp_path / "ck")

    def run(n_iters, port):
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        ps = [ctx.Process(target=_dist_ck_worker,
                          args=(r, 2, port, ckdir, n_iters, q))
              for r in range(2)]
        for p in ps:
            p.start()
        res = {}
        for _ in range(2):
            r, s = q.get(timeout=200)
            res[r] = s
        for p in ps:
            p.join(timeout=30)
        assert not any(str(s).startswith("ERROR") for s in res.values()), res
        return res

    res4 = run(4, 29961)      # "crashes" after 4 iterations (checkpointed)
    assert res4[0] == res4[1]
    from mmlspark_amd.models.gbdt.trainer import load_checkpoint
    b_ck, it = load_checkpoint(ckdir)
    assert it == 4 and b_ck.num_trees == 4

    res8 = run(8, 29962)      # gang restart resumes 5..8
    assert res8[0] == res8[1]
    from mmlspark_amd.models.gbdt.booster import Booster
    b = Booster.load_from_string(res8[0])
    assert b.num_trees == 8
    # the first 4 trees are exactly the pre-crash trees
    import json as _json
    t_resumed = _json.loads(res8[0])["trees"][:4]
    t_before = _json.loads(res4[0])["trees"]
    assert t_resumed == t_before


def _worker_resume_rank0_only(rank, world, port, q, ckdir):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm
        rng = np.random.default_rng(0)
        X = rng.normal(size=(2000, 8)).astype(np.float32)
        y = (X[:, 0] > 0).astype(np.float32)
        sl = slice(rank * 1000, (rank + 1) * 1000)
        cfg = TrainConfig(num_iterations=6, num_leaves=7, seed=3)
        # only rank 0 can see the checkpoint dir (node-local disk shape)
        ck = ckdir if rank == 0 else os.path.join(ckdir, "nonexistent")
        booster, _ = train_booster(torch.from_numpy(X[sl]),
                                   torch.from_numpy(y[sl]), cfg,
                                   make_objective("binary"), Comm(),
                                   checkpoint_dir=ck, checkpoint_every=2)
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


@pytest.mark.timeout(240)
def test_resume_with_rank0_only_checkpoint(tmp_path):
    """Rank 0 holds a 4-iteration checkpoint, rank 1 sees nothing (node-
    local disks): rank 0's resume state must be broadcast so both ranks
    start at the same iteration instead of deadlocking on mismatched
    collective counts, and end with identical boosters."""
    import multiprocessing as mp
    from mmlspark_amd.models.gbdt.objectives import make_objective
    from mmlspark_amd.models.gbdt.trainer import (TrainConfig, _save_checkpoint,
                                                  train_booster)
    from mmlspark_amd.parallel.comm import Comm
    ckdir = str(tmp_path / "ck")
    # seed the checkpoint: a 4-iteration single-rank run on the SAME data
    rng = np.random.default_rng(0)
    X = rng.normal(size=(2000, 8)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    pre, _ = train_booster(torch.from_numpy(X[:1000]),
                           torch.from_numpy(y[:1000]),
                           TrainConfig(num_iterations=4, num_leaves=7, seed=3),
                           make_objective("binary"), Comm())
    _save_checkpoint(ckdir, pre, 4)

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29961
    procs = [ctx.Process(target=_worker_resume_rank0_only,
                         args=(r, 2, port, q, ckdir)) for r in range(2)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(2):
        rank, s = q.get(timeout=180)
        results[rank] = s
    for p in procs:
        p.join(timeout=30)
    assert not any(str(s).startswith("ERROR") for s in results.values()), \
        results
    assert results[0] == results[1]
    import json as _json
    assert len(_json.loads(results[0])["trees"]) == 6  # 4 resumed + 2 new
