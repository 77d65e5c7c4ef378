"""HIP kernel numerics vs the torch fp32 CPU reference (run on MI355X)."""
import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs ROCm GPU")


@requires_gpu
def test_ext_is_loaded():
    from mmlspark_amd.ops.backend import hip_available
    assert hip_available(), "HIP extension must be importable on the GPU box"


@requires_gpu
def test_hist_build_matches_cpu():
    from mmlspark_amd.ops import backend, cpu_ref
    g = torch.Generator().manual_seed(0)
    n, nf, nb = 100_000, 26, 255   # nf not multiple of 4 → padding path
    ngroups = (nf + 3) // 4
    binned = torch.randint(0, nb, (ngroups, n, 4), generator=g,
                           dtype=torch.uint8)
    rows = torch.randperm(n, generator=g)[: n // 3].to(torch.int32).sort().values
    grad = torch.randn(n, generator=g)
    hess = torch.rand(n, generator=g) + 0.1
    ref = cpu_ref.hist_build(binned, rows, grad, hess, nb)
    out = backend.hist_build(binned.cuda(), rows.cuda(), grad.cuda(),
                             hess.cuda(), nb).cpu()
    assert out.shape == ref.shape
    assert torch.allclose(out[:, :, 2], ref[:, :, 2])          # counts exact
    assert torch.allclose(out[:, :, 0], ref[:, :, 0], atol=2e-3, rtol=1e-4)
    assert torch.allclose(out[:, :, 1], ref[:, :, 1], atol=2e-3, rtol=1e-4)


@requires_gpu
def test_bin_matrix_matches_cpu():
    from mmlspark_amd.ops import backend, cpu_ref
    g = torch.Generator().manual_seed(1)
    n, nf, nb = 50_000, 10, 255
    X = torch.randn(n, nf, generator=g)
    X[::97, 3] = float("nan")
    ub = torch.sort(torch.randn(nf, nb - 1, generator=g), dim=1).values
    ub[:, -1] = float("inf")
    ref = cpu_ref.bin_matrix(X, ub, nb)
    out = backend.bin_matrix(X.cuda(), ub.cuda(), nb).cpu()
    assert torch.equal(ref, out)


@requires_gpu
def test_predict_forest_matches_cpu(binary_df):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    m = LightGBMClassifier(numIterations=10, numLeaves=15,
                           device="cpu").fit(binary_df)
    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy()))
    raw_cpu = m.booster.predict_raw(X)
    m.booster.invalidate_cache()
    raw_gpu = m.booster.predict_raw(X.cuda()).cpu()
    assert torch.allclose(raw_cpu, raw_gpu, atol=1e-5)
    leaf_cpu = m.booster.predict_leaf(X)
    leaf_gpu = m.booster.predict_leaf(X.cuda()).cpu()
    assert torch.equal(leaf_cpu, leaf_gpu)


@requires_gpu
def test_gpu_training_end_to_end(binary_df):
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    m = LightGBMClassifier(numIterations=30, numLeaves=15, learningRate=0.2,
                           device="cuda").fit(binary_df)
    out = m.transform(binary_df)
    prob = np.stack(out["probability"].to_numpy())[:, 1]
    y = binary_df["label"].to_numpy()
    assert roc_auc_score(y, prob) > 0.95


@requires_gpu
def test_gpu_cpu_training_parity(binary_df):
    """Same data, same params: GPU-trained and CPU-trained boosters agree
    closely (same algorithm, same binning; fp32 atomics reorder sums)."""
    from sklearn.metrics import roc_auc_score
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    y = binary_df["label"].to_numpy()
    aucs = []
    for dev in ("cpu", "cuda"):
        m = LightGBMClassifier(numIterations=15, numLeaves=15, device=dev).fit(binary_df)
        prob = np.stack(m.transform(binary_df)["probability"].to_numpy())[:, 1]
        aucs.append(roc_auc_score(y, prob))
    assert abs(aucs[0] - aucs[1]) < 0.01, aucs


@requires_gpu
def test_vw_kernels():
    from mmlspark_amd.ops import backend
    g = torch.Generator().manual_seed(2)
    bits = 18
    tbl = 1 << bits
    n_ex, feats_per = 20_000, 30
    idx = torch.randint(0, tbl, (n_ex * feats_per,), generator=g,
                        dtype=torch.int32)
    val = torch.randn(n_ex * feats_per, generator=g)
    off = torch.arange(0, n_ex + 1, dtype=torch.int64) * feats_per
    w_true = torch.randn(tbl, generator=g) * 0.1
    # labels from a sparse linear model
    labels = torch.zeros(n_ex)
    for s in range(0, n_ex, 4096):
        e = min(s + 4096, n_ex)
        for i in range(s, e):
            sl = slice(int(off[i]), int(off[i + 1]))
            labels[i] = torch.sign((w_true[idx[sl].long()] * val[sl]).sum())
    labels[labels == 0] = 1.0

    w = torch.zeros(tbl).cuda()
    gacc = torch.zeros(tbl).cuda()
    idx_d, val_d, off_d, y_d = idx.cuda(), val.cuda(), off.cuda(), labels.cuda()
    for _ in range(4):
        backend.vw_sgd_minibatch(idx_d, val_d, off_d, y_d, w, gacc,
                                 0.5, 0.0, 0.5, "logistic")
    preds = backend.vw_predict(idx_d, val_d, off_d, w).cpu()
    acc = ((preds.sign() == labels).float().mean())
    # hogwild atomic order varies run-to-run: bar leaves slack below the
    # typical ~0.9
    assert float(acc) > 0.75, float(acc)

    # --invariant path: with DISJOINT per-example index sets (no hash
    # collisions) hogwild order is irrelevant, so GPU must match the CPU
    # reference tightly
    from mmlspark_amd.models.vw import sgd_ref
    n2, fp2 = 8000, 30
    idx2 = torch.arange(n2 * fp2, dtype=torch.int32) % tbl
    val2 = torch.randn(n2 * fp2, generator=g)
    off2 = torch.arange(0, n2 + 1, dtype=torch.int64) * fp2
    y2 = torch.where(torch.rand(n2, generator=g) > 0.5, 1.0, -1.0)
    hw = (torch.rand(n2, generator=g) * 100).clamp_min(0.5)
    wg = torch.zeros(tbl).cuda()
    gg = torch.zeros(tbl).cuda()
    backend.vw_sgd_minibatch(idx2.cuda(), val2.cuda(), off2.cuda(), y2.cuda(),
                             wg, gg, 0.5, 0.0, 0.5, "logistic", hw.cuda(),
                             None, True)
    wc = torch.zeros(tbl)
    gc2 = torch.zeros(tbl)
    sgd_ref.vw_sgd_minibatch(idx2, val2, off2, y2, wc, gc2, 0.5, 0.0,
                             0.5, "logistic", ex_weight=hw, invariant=True)
    pg = backend.vw_predict(idx2.cuda(), val2.cuda(), off2.cuda(), wg).cpu()
    pc = sgd_ref.vw_predict(idx2, val2, off2, wc)
    assert torch.isfinite(pg).all()
    err = (pg - pc).abs().mean() / pc.abs().mean().clamp_min(1e-6)
    assert float(err) < 0.02, float(err)


@requires_gpu
def test_split_scan_matches_cpu():
    from mmlspark_amd.ops import backend, cpu_ref
    g = torch.Generator().manual_seed(3)
    nh, nf_pad, nb = 2, 28, 255
    hists = torch.zeros(nh, nf_pad, nb, 3)
    hists[:, :, :, 0] = torch.randn(nh, nf_pad, nb, generator=g)
    hists[:, :, :, 1] = torch.rand(nh, nf_pad, nb, generator=g) + 0.01
    hists[:, :, :, 2] = torch.randint(0, 50, (nh, nf_pad, nb), generator=g).float()
    mask = torch.ones(nf_pad, dtype=torch.bool)
    mask[5] = False
    for fm in (None, mask):
        ref = cpu_ref.split_scan(hists, nb, 0.1, 0.5, 3.0, 1e-3, 0.0, 26, fm)
        out = backend.split_scan(hists.cuda(), nb, 0.1, 0.5, 3.0, 1e-3, 0.0,
                                 26, fm.cuda() if fm is not None else None).cpu()
        # gains equal within fp tolerance; chosen split identical
        assert torch.allclose(ref[:, 0], out[:, 0], rtol=1e-3, atol=1e-3), (ref, out)
        assert torch.equal(ref[:, 1:3], out[:, 1:3]), (ref, out)
        assert torch.allclose(ref[:, 3:], out[:, 3:], rtol=1e-3, atol=1e-2)


@requires_gpu
def test_hist_build_fixed_matches_cpu():
    from mmlspark_amd.ops import backend, cpu_ref
    g = torch.Generator().manual_seed(5)
    n, nf, nb = 200_000, 26, 255
    ngroups = (nf + 3) // 4
    binned = torch.randint(0, nb, (ngroups, n, 4), generator=g,
                           dtype=torch.uint8)
    rows = torch.randperm(n, generator=g)[: n // 2].to(torch.int32).sort().values
    grad = torch.randn(n, generator=g)
    hess = torch.rand(n, generator=g) + 0.01
    ref = cpu_ref.hist_build(binned, rows, grad, hess, nb)
    gmax = float(grad.abs().max())
    hmax = float(hess.max())
    sg = (2.0 ** 61) / (n * gmax)
    sh = (2.0 ** 24) / hmax
    out = backend.hist_build_fixed(binned.cuda(), rows.cuda(), grad.cuda(),
                                   hess.cuda(), nb, sg, sh).cpu()
    outf = out.double() * torch.tensor([1.0 / sg, 1.0 / sh, 1.0]).double()
    assert torch.equal(outf[:, :, 2].float(), ref[:, :, 2])  # counts exact
    assert torch.allclose(outf[:, :, 0].float(), ref[:, :, 0], atol=2e-3,
                          rtol=1e-4)
    assert torch.allclose(outf[:, :, 1].float(), ref[:, :, 1], atol=2e-3,
                          rtol=1e-4)


@requires_gpu
def test_partition_rows_kernel_matches_cpu():
    from mmlspark_amd.ops import backend, cpu_ref
    g = torch.Generator().manual_seed(6)
    n, nf, nb = 300_000, 12, 255
    ngroups = (nf + 3) // 4
    binned = torch.randint(0, nb, (ngroups, n, 4), generator=g,
                           dtype=torch.uint8)
    rows = torch.randperm(n, generator=g)[: n // 2].to(torch.int32).sort().values
    for feature, thr in ((0, 100), (5, 3), (11, 254)):
        l_ref, r_ref = cpu_ref.partition_rows(binned, rows, feature, thr)
        l_gpu, r_gpu = backend.partition_rows(binned.cuda(), rows.cuda(),
                                              feature, thr)
        assert torch.equal(l_ref, l_gpu.cpu()), feature
        assert torch.equal(r_ref, r_gpu.cpu()), feature


@requires_gpu
def test_categorical_gpu_matches_cpu():
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(13)
    n = 4000
    cat = rng.integers(0, 12, size=n).astype(np.float32)
    noise = rng.normal(size=(n, 3)).astype(np.float32)
    good = {1, 4, 7, 10}
    y = np.array([1.0 if int(c) in good else 0.0 for c in cat], np.float32)
    X = np.column_stack([cat, noise]).astype(np.float32)
    import pandas as pd
    df = pd.DataFrame({"features": list(X), "label": y})
    m = LightGBMClassifier(numIterations=10, numLeaves=15,
                           categoricalSlotIndexes=[0], minDataInLeaf=5,
                           device="cuda").fit(df)
    out = m.transform(df)
    acc = (out["prediction"].to_numpy() == y).mean()
    assert acc > 0.98, acc
    # GPU and CPU predict paths agree on categorical trees
    Xt = torch.from_numpy(X)
    raw_gpu = m.booster.predict_raw(Xt.cuda()).cpu()
    m.booster.invalidate_cache()
    raw_cpu = m.booster.predict_raw(Xt)
    assert torch.allclose(raw_gpu, raw_cpu, atol=1e-5)


@requires_gpu
def test_image_featurizer_gpu():
    import pandas as pd
    from mmlspark_amd.models.image_featurizer import ImageFeaturizer
    rng = np.random.default_rng(20)
    imgs = [rng.integers(0, 255, size=(32, 32, 3)).astype(np.uint8)
            for _ in range(8)]
    df = pd.DataFrame({"image": imgs})
    f = ImageFeaturizer(modelName="ResNet50", imageSize=64, cutOutputLayers=1,
                        device="cuda", batchSize=4)
    feats = np.stack(f.transform(df)["features"].to_numpy())
    assert feats.shape == (8, 2048)
    assert np.isfinite(feats).all()


@requires_gpu
def test_tabular_shap_gpu_model(binary_df):
    import pandas as pd
    from mmlspark_amd.explainers.shap import TabularSHAP
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    X = np.stack(binary_df["features"].to_numpy())
    cols = [f"c{i}" for i in range(X.shape[1])]
    df = pd.DataFrame(X, columns=cols)
    df["label"] = binary_df["label"].to_numpy()
    model = LightGBMClassifier(featureCols=cols, numIterations=10,
                               numLeaves=15, device="cuda").fit(df)
    shap = TabularSHAP(inputCols=cols, model=model, targetCol="probability",
                       targetClasses=[1], numSamples=128,
                       backgroundData=df.head(100))
    out = shap.transform(df.head(4))
    exp = np.stack(out["explanation"].to_numpy())
    probs = np.stack(model.transform(df.head(4))["probability"].to_numpy())[:, 1]
    np.testing.assert_allclose(exp[:, 0, :].sum(axis=1), probs, atol=1e-3)


@requires_gpu
def test_serving_gpu_hipgraph(binary_df):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    from mmlspark_amd.serving.server import LowLatencyGBDTScorer
    model = LightGBMClassifier(numIterations=20, numLeaves=15,
                               device="cuda").fit(binary_df)
    scorer = LowLatencyGBDTScorer(model.booster, max_batch=4)
    assert scorer.graph is not None, "hipGraph capture must succeed on GPU"
    X = np.stack(binary_df["features"].to_numpy()[:3])
    p = scorer.score(X)
    ref = np.stack(model.transform(binary_df.head(3))["probability"]
                   .to_numpy())[:, 1]
    np.testing.assert_allclose(p[:, 0], ref, atol=1e-5)


@requires_gpu
def test_native_grower_distributed_codepath(binary_df):
    """Exercise the multi-rank branches of the C++ grower on one GPU (pg=None
    so the reduce is skipped but every distributed branch — device-side
    partition counts, pinned nl readback, speculation — runs)."""
    from mmlspark_amd.models.gbdt.objectives import make_objective
    from mmlspark_amd.models.gbdt.trainer import (TrainConfig, TrainingSession)
    from mmlspark_amd.parallel.comm import Comm
    from mmlspark_amd.ops import _hip_grower

    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy())).cuda()
    y = torch.from_numpy(binary_df["label"].to_numpy()).float().cuda()
    cfg = TrainConfig(num_iterations=1, num_leaves=15)
    ses = TrainingSession(X, y, cfg, make_objective("binary"), Comm())
    g, h = ses.objective.grad_hess(ses.preds, y, None)
    grower = ses.grower
    grower.set_scales(g[:, 0], h[:, 0])
    d = _hip_grower.grow_tree_native(
        grower.binned, grower.binned_pair, ses.all_rows, g[:, 0].contiguous(),
        h[:, 0].contiguous(), cfg.max_bin, grower.nf, grower.scale_g,
        grower.scale_h, 0.0, 0.0, float(cfg.min_data_in_leaf), 1e-3, 0.0,
        0.0, 15, -1, None, None, 10.0, None, True)
    feature = d["feature"].numpy()
    assert (feature >= 0).sum() == 14  # 15 leaves → 14 internal nodes
    # leaf segments partition all rows exactly
    assert int(d["leaf_offsets"][-1]) == len(binary_df)


@requires_gpu
def test_tree_shap_gpu_matches_cpu(binary_df):
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    m = LightGBMClassifier(numIterations=10, numLeaves=15,
                           device="cpu").fit(binary_df)
    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy()[:64]))
    cpu = m.booster.predict_contrib(X)
    gpu = m.booster.predict_contrib(X.cuda())
    np.testing.assert_allclose(gpu, cpu, atol=2e-3, rtol=1e-3)
    # additivity through the GPU path
    raw = m.booster.predict_raw(X).squeeze(-1).numpy()
    np.testing.assert_allclose(gpu.sum(axis=1), raw, atol=2e-3)


@requires_gpu
def test_contextual_bandit_gpu():
    import pandas as pd
    from mmlspark_amd.core.schema import SparseVector
    from mmlspark_amd.models.vw.estimators import VowpalWabbitContextualBandit
    rng = np.random.default_rng(2)
    size = 1 << 14
    rows = []
    for _ in range(1500):
        ctx = int(rng.integers(0, 3))
        rows.append({
            "shared": SparseVector(size, [100 + ctx], [1.0]),
            "features": [SparseVector(size, [2000 + a], [1.0])
                         for a in range(3)],
            "chosenAction": int(rng.integers(0, 3)) + 1,
            "cost": 0.0, "probability": 1 / 3, "ctx": ctx})
        rows[-1]["cost"] = 0.0 if rows[-1]["chosenAction"] - 1 == ctx else 1.0
    df = pd.DataFrame(rows)
    cb = VowpalWabbitContextualBandit(numPasses=6, numBits=14,
                                      learningRate=0.5, device="cuda").fit(df)
    out = cb.transform(df)
    acc = ((out["prediction"].to_numpy() - 1) == df["ctx"].to_numpy()).mean()
    assert acc > 0.85  # hogwild order varies run-to-run


@requires_gpu
def test_tree_shap_gpu_multiclass():
    import pandas as pd
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(4)
    X = rng.normal(size=(1500, 6)).astype(np.float32)
    y = np.digitize(X[:, 0] + X[:, 1], [-1.0, 1.0]).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = LightGBMClassifier(objective="multiclass", numIterations=8,
                           numLeaves=7, device="cpu").fit(df)
    Xt = torch.from_numpy(X[:20])
    cpu = m.booster.predict_contrib(Xt)
    gpu = m.booster.predict_contrib(Xt.cuda())
    np.testing.assert_allclose(gpu, cpu, atol=2e-3, rtol=1e-2)


@requires_gpu
def test_native_grower_device_count_path(binary_df):
    """MMLSPARK_AMD_FORCE_DIST_GROWER=1 forces the distributed grower code
    path (device-side partition counts + pinned nl readback + speculation)
    on one rank; the model must match the single-rank fast path exactly."""
    import os
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    df = binary_df
    m1 = LightGBMClassifier(numIterations=8, numLeaves=31,
                            device="cuda").fit(df)
    os.environ["MMLSPARK_AMD_FORCE_DIST_GROWER"] = "1"
    try:
        m2 = LightGBMClassifier(numIterations=8, numLeaves=31,
                                device="cuda").fit(df)
    finally:
        del os.environ["MMLSPARK_AMD_FORCE_DIST_GROWER"]
    X = torch.from_numpy(np.stack(df["features"].to_numpy()[:500])).cuda()
    p1 = m1.booster.predict_raw(X).cpu()
    p2 = m2.booster.predict_raw(X).cpu()
    assert torch.equal(p1, p2), float((p1 - p2).abs().max())


@requires_gpu
def test_hist_build_fixed_pair_matches_unpaired():
    """Paired-plane histogram == unpaired fixed-point histogram exactly
    (same fixed-point integers, one 8B load per row per block)."""
    from mmlspark_amd.ops import backend
    g = torch.Generator().manual_seed(13)
    n, nf, nb = 300_000, 26, 255  # odd group count → zero-padded high plane
    ngroups = (nf + 3) // 4
    binned = torch.randint(0, nb, (ngroups, n, 4), generator=g,
                           dtype=torch.uint8).cuda()
    g32 = binned.view(torch.int32).reshape(ngroups, -1)
    if ngroups % 2:
        g32 = torch.cat([g32, torch.zeros_like(g32[:1])])
    lo = g32[0::2].to(torch.int64) & 0xFFFFFFFF
    hi = g32[1::2].to(torch.int64) & 0xFFFFFFFF
    pair = (lo | (hi << 32)).contiguous()
    rows = torch.randperm(n, generator=g)[: n // 4].to(
        torch.int32).sort().values.cuda()
    grad = torch.randn(n, generator=g).cuda()
    hess = (torch.rand(n, generator=g) + 0.1).cuda()
    sg, sh = 2.0 ** 40 / n, 2.0 ** 24
    nf_pad = ngroups * 4
    tail = nf_pad - (pair.shape[0] - 1) * 8
    ref = backend.hist_build_fixed(binned, rows, grad, hess, nb, sg, sh)
    out = backend.hist_build_fixed_pair(pair, rows, grad, hess, nb, tail,
                                        sg, sh)
    assert torch.equal(out[:nf_pad].cpu(), ref.cpu())
    # zero-pad features are skipped entirely (no bin-0 atomic hotspot)
    if out.shape[0] > nf_pad:
        assert int(out[nf_pad:].abs().sum()) == 0


@requires_gpu
def test_vw_bfgs_gpu(binary_df):
    """--bfgs full-batch L-BFGS path on device tensors (differentiable
    gather/segment-sum forward under torch L-BFGS on ROCm)."""
    from mmlspark_amd.models.vw.estimators import VowpalWabbitClassifier
    m = VowpalWabbitClassifier(bfgs=True, lossFunction="logistic",
                               device="cuda").fit(binary_df)
    y = binary_df["label"].to_numpy()
    acc = (m.transform(binary_df)["prediction"].to_numpy() == y).mean()
    assert acc > 0.9, acc


@requires_gpu
def test_categorical_native_grower_bit_identical():
    """categorical one-vs-rest splits now run inside the
    native arena grower (bitset partition + host cat scan) and must produce
    BIT-IDENTICAL boosters to the Python grower on the same GPU data."""
    import os
    import pandas as pd
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(17)
    n = 20_000
    cat1 = rng.integers(0, 24, size=n).astype(np.float32)
    cat2 = rng.integers(0, 6, size=n).astype(np.float32)
    num = rng.normal(size=(n, 4)).astype(np.float32)
    good = {2, 5, 9, 13, 20}
    y = ((np.isin(cat1.astype(int), list(good))) ^ (num[:, 0] > 0.7)
         ).astype(np.float32)
    X = np.column_stack([cat1, num[:, :2], cat2, num[:, 2:]]).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    kw = dict(numIterations=12, numLeaves=31, categoricalSlotIndexes=[0, 3],
              minDataInLeaf=5, featureFraction=0.8, device="cuda")
    m_native = LightGBMClassifier(**kw).fit(df)
    os.environ["MMLSPARK_AMD_NO_NATIVE_GROWER"] = "1"
    try:
        m_py = LightGBMClassifier(**kw).fit(df)
    finally:
        del os.environ["MMLSPARK_AMD_NO_NATIVE_GROWER"]
    s_native = m_native.booster.save_to_string()
    s_py = m_py.booster.save_to_string()
    assert any((t.cat_offset >= 0).any() for t in m_native.booster.trees), \
        "native grower must actually take categorical splits"
    assert s_native == s_py


@requires_gpu
def test_multiclass_native_grower_end_to_end():
    """Multiclass (K trees per iteration) through the native arena grower:
    accuracy on a 4-class task + CPU/GPU save-string structural agreement."""
    import pandas as pd
    from sklearn.metrics import accuracy_score
    from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
    rng = np.random.default_rng(31)
    n, nf, K = 20_000, 12, 4
    X = rng.normal(size=(n, nf)).astype(np.float32)
    centers = rng.normal(size=(K, nf)) * 2
    y = np.argmin(((X[:, None, :] - centers[None]) ** 2).sum(-1),
                  axis=1).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = LightGBMClassifier(numIterations=15, numLeaves=31,
                           objective="multiclass", device="cuda").fit(df)
    assert m.booster.num_trees == 15 * K
    out = m.transform(df)
    acc = accuracy_score(y, out["prediction"].to_numpy())
    assert acc > 0.9, acc
    prob = np.stack(out["probability"].to_numpy())
    assert prob.shape == (n, K)
    np.testing.assert_allclose(prob.sum(axis=1), 1.0, atol=1e-5)
