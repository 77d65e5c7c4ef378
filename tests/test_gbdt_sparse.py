"""Sparse CSR GBDT ingestion (CPU): accuracy vs the dense path, memory-shape
sanity, estimator auto-detect — the parity of LGBM_DatasetCreateFromCSR
(DatasetAggregator.scala:442) + sparse auto-detect (DatasetUtils.scala:49)."""
import numpy as np
import pandas as pd
import pytest
import torch

from mmlspark_amd.core.schema import SparseVector
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
from mmlspark_amd.models.gbdt.sparse import (CsrMatrix, bin_csr,
                                             fit_bin_mapper_csr, looks_sparse)


def _sparse_data(seed=0, n=4000, nf=50, nnz=8):
    """Synthetic sparse binary task: label from a few informative features."""
    rng = np.random.default_rng(seed)
    rows = []
    y = np.zeros(n, dtype=np.float32)
    w = rng.normal(size=nf)
    for i in range(n):
        idx = np.sort(rng.choice(nf, size=nnz, replace=False)).astype(np.int32)
        val = rng.normal(size=nnz).astype(np.float32)
        s = float((w[idx] * val).sum())
        y[i] = 1.0 if s + rng.normal() * 0.3 > 0 else 0.0
        rows.append(SparseVector(nf, idx, val))
    return rows, y


def _densify(rows, nf):
    X = np.zeros((len(rows), nf), dtype=np.float32)
    for i, v in enumerate(rows):
        X[i, v.indices] = v.values
    return X


def test_looks_sparse_detection():
    rows, _ = _sparse_data(n=20)
    assert looks_sparse(pd.Series(rows))
    assert not looks_sparse(pd.Series([np.zeros(5), np.ones(5)]))


def test_csr_matrix_roundtrip():
    rows, _ = _sparse_data(n=100, nf=30)
    m = CsrMatrix.from_sparse_vectors(pd.Series(rows))
    assert m.shape == (100, 30)
    d = m.densify().numpy()
    np.testing.assert_allclose(d, _densify(rows, 30))
    # row_slice keeps content
    d2 = m.row_slice(10, 40).densify().numpy()
    np.testing.assert_allclose(d2, d[10:40])


def test_sparse_binning_matches_dense_rule():
    """zero_bin and entry bins must follow the dense searchsorted rule so
    thresholds exported at predict time stay consistent."""
    rows, _ = _sparse_data(seed=3, n=2000, nf=20)
    csr = CsrMatrix.from_sparse_vectors(pd.Series(rows))
    mapper = fit_bin_mapper_csr(csr, n_bins=63, sample_size=2000)
    shard = bin_csr(csr, mapper)
    ub = mapper.upper_bounds
    # every stored entry: bin == searchsorted(bounds, value)
    for e in range(0, csr.nnz, 97):
        f = int(csr.col[e])
        v = float(csr.val[e])
        expect = int(torch.searchsorted(ub[f], torch.tensor(v), right=False))
        assert int(shard.binv[e]) == min(expect, 62)
    for f in range(20):
        expect = int(torch.searchsorted(ub[f], torch.tensor(0.0), right=False))
        assert int(shard.zero_bin[f]) == min(expect, 62)


def test_sparse_hist_zero_correction_totals():
    """After implicit-zero correction every feature's histogram sums to the
    exact leaf totals (feature 0 'owns every row' invariant)."""
    from mmlspark_amd.models.gbdt.trainer import (SparseTreeGrower,
                                                  TrainConfig, TrainingStats)
    from mmlspark_amd.parallel.comm import Comm
    rows, y = _sparse_data(seed=1, n=1000, nf=15)
    csr = CsrMatrix.from_sparse_vectors(pd.Series(rows))
    cfg = TrainConfig(max_bin=31)
    mapper = fit_bin_mapper_csr(csr, n_bins=31, sample_size=1000)
    shard = bin_csr(csr, mapper)
    g = SparseTreeGrower(shard, 15, cfg, Comm(), TrainingStats(), mapper)
    grad = torch.randn(1000)
    hess = torch.rand(1000) + 0.1
    g.set_scales(grad, hess)
    sel = torch.arange(0, 1000, 3, dtype=torch.int32)
    h = g._hist(sel, grad, hess)
    sums = h.sum(dim=1)  # (nf, 3)
    for f in range(15):
        assert torch.equal(sums[f], sums[0])
    assert int(sums[0][2]) == sel.numel()


def test_sparse_vs_dense_training_quality():
    """Sparse CSR training must match dense-path quality on the same data
    (binning differs slightly — mixture quantiles — so compare AUC)."""
    from sklearn.metrics import roc_auc_score
    rows, y = _sparse_data(seed=5, n=5000, nf=40, nnz=10)
    df_sp = pd.DataFrame({"features": rows, "label": y})
    X = _densify(rows, 40)
    df_dn = pd.DataFrame({"features": list(X), "label": y})

    m_sp = LightGBMClassifier(numIterations=30, numLeaves=15,
                              learningRate=0.2).fit(df_sp)
    m_dn = LightGBMClassifier(numIterations=30, numLeaves=15,
                              learningRate=0.2).fit(df_dn)
    p_sp = np.stack(m_sp.transform(df_sp)["probability"].to_numpy())[:, 1]
    p_dn = np.stack(m_dn.transform(df_dn)["probability"].to_numpy())[:, 1]
    a_sp = roc_auc_score(y, p_sp)
    a_dn = roc_auc_score(y, p_dn)
    assert a_sp > 0.9, a_sp
    assert a_sp > a_dn - 0.03, (a_sp, a_dn)


def test_sparse_matrix_type_forced_and_save_load(tmp_path):
    rows, y = _sparse_data(seed=7, n=2000, nf=30)
    df = pd.DataFrame({"features": rows, "label": y})
    m = LightGBMClassifier(numIterations=10, numLeaves=15,
                           matrixType="sparse").fit(df)
    out = m.transform(df)
    assert {"prediction", "probability", "rawPrediction"} <= set(out.columns)
    acc = (out["prediction"].to_numpy() == y).mean()
    assert acc > 0.8, acc
    # persistence round trip scores sparse input identically
    p = str(tmp_path / "m")
    m.save(p)
    from mmlspark_amd.core.serialize import load_stage
    m2 = load_stage(p)
    out2 = m2.transform(df)
    np.testing.assert_allclose(
        np.stack(out["probability"].to_numpy()),
        np.stack(out2["probability"].to_numpy()), atol=1e-6)


def test_sparse_high_dim_memory_shape():
    """1k×100k sparse: CSR storage is O(nnz); densified this would be 400 MB.
    (The 1M×100k scale run is the GPU-marked test.)"""
    rng = np.random.default_rng(11)
    n, nf, nnz = 1000, 100_000, 20
    rows = []
    for i in range(n):
        idx = np.sort(rng.choice(nf, size=nnz, replace=False))
        rows.append(SparseVector(nf, idx, rng.normal(size=nnz)))
    w_idx = rng.choice(nf, size=50, replace=False)
    y = np.array([1.0 if np.intersect1d(v.indices, w_idx).size else 0.0
                  for v in rows], dtype=np.float32)
    df = pd.DataFrame({"features": rows, "label": y})
    m = LightGBMClassifier(numIterations=3, numLeaves=7, maxBin=15,
                           minDataInLeaf=5).fit(df)
    assert m.booster.num_trees == 3
    assert m.booster.n_features == nf


def test_sparse_distributed_gloo_bit_exact():
    """2-rank gloo: sparse histograms are fixed-point int64, so synchronized
    growth must produce identical boosters on every rank."""
    import torch.multiprocessing as mp

    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_sparse_dist_worker, args=(r, 2, 29876, q))
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
    assert results[0] == results[1]


def _sparse_dist_worker(rank, world, port, q):
    try:
        import os
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            "gloo", init_method=f"tcp://127.0.0.1:{port}",
            rank=rank, world_size=world)
        from mmlspark_amd.models.gbdt.objectives import make_objective
        from mmlspark_amd.models.gbdt.trainer import TrainConfig, train_booster
        from mmlspark_amd.parallel.comm import Comm

        rows, y = _sparse_data(seed=2, n=3000, nf=25)
        n_shard = len(rows) // world
        sl = slice(rank * n_shard, (rank + 1) * n_shard)
        csr = CsrMatrix.from_sparse_vectors(pd.Series(rows[sl]))
        yt = torch.from_numpy(y[sl])
        booster, _ = train_booster(csr, yt,
                                   TrainConfig(num_iterations=8, num_leaves=15,
                                               seed=3),
                                   make_objective("binary"), Comm())
        q.put((rank, booster.save_to_string()))
        torch.distributed.destroy_process_group()
    except Exception as e:  # pragma: no cover
        q.put((rank, f"ERROR: {e!r}"))


def test_sparse_property_random_shapes():
    """Property sweep: random sparse shapes/params must train, score,
    save/load without error (empty rows, single-feature, tiny data,
    unsorted indices, duplicate-free)."""
    rng = np.random.default_rng(23)
    for trial in range(6):
        n = int(rng.integers(30, 400))
        nf = int(rng.integers(2, 200))
        rows = []
        for i in range(n):
            k = int(rng.integers(0, min(nf, 12) + 1))  # may be 0 (empty row)
            idx = rng.choice(nf, size=k, replace=False).astype(np.int32)
            rng.shuffle(idx)  # unsorted on purpose — ingestion must sort
            rows.append(SparseVector(nf, idx,
                                     rng.normal(size=k).astype(np.float32)))
        y = rng.integers(0, 2, size=n).astype(np.float32)
        df = pd.DataFrame({"features": rows, "label": y})
        m = LightGBMClassifier(numIterations=3,
                               numLeaves=int(rng.integers(2, 15)),
                               maxBin=int(rng.integers(4, 64)),
                               minDataInLeaf=2).fit(df)
        out = m.transform(df)
        assert len(out) == n
        p = np.stack(out["probability"].to_numpy())
        assert np.isfinite(p).all(), trial
        b2 = m.booster.load_from_string(m.booster.save_to_string())
        assert b2.num_trees == m.booster.num_trees


def test_sparse_weight_and_validation_cols():
    """weightCol + validationIndicatorCol flow through the sparse path."""
    rng = np.random.default_rng(29)
    n, nf = 1500, 40
    rows, y = [], np.zeros(n, dtype=np.float32)
    w = rng.normal(size=nf)
    for i in range(n):
        idx = np.sort(rng.choice(nf, size=8, replace=False)).astype(np.int32)
        val = rng.normal(size=8).astype(np.float32)
        y[i] = 1.0 if (w[idx] * val).sum() > 0 else 0.0
        rows.append(SparseVector(nf, idx, val))
    df = pd.DataFrame({"features": rows, "label": y,
                       "wt": rng.random(n).astype(np.float32) + 0.5,
                       "isVal": rng.random(n) < 0.25})
    m = LightGBMClassifier(numIterations=25, numLeaves=15, weightCol="wt",
                           validationIndicatorCol="isVal", metric="auc",
                           earlyStoppingRound=5).fit(df)
    assert m.booster.num_trees >= 1
    evals = m._training_stats.evals
    assert evals and "auc" in evals[0]["valid_0"]
