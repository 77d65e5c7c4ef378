"""VW-equivalent module: murmur parity, featurizer, SGD learners, CB."""
import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.core.schema import SparseVector
from mmlspark_amd.models.vw.estimators import (
    VowpalWabbitClassifier, VowpalWabbitContextualBandit,
    VowpalWabbitRegressor)
from mmlspark_amd.models.vw.featurizer import (VowpalWabbitFeaturizer,
                                               VowpalWabbitInteractions)
from mmlspark_amd.models.vw.murmur import hash_string, murmur3_32


def test_murmur3_known_vectors():
    # canonical MurmurHash3_x86_32 test vectors
    assert murmur3_32(b"", 0) == 0
    assert murmur3_32(b"", 1) == 0x514E28B7
    assert murmur3_32(b"hello", 0) == 0x248BFA47
    assert murmur3_32(b"hello, world", 0) == 0x149BBB7F
    assert murmur3_32(b"The quick brown fox jumps over the lazy dog", 0) == 0x2E4FF723


def test_featurizer_basic():
    df = pd.DataFrame({
        "num": [1.5, 0.0, 2.0],
        "cat": ["a", "b", "a"],
        "txt": ["hello world", "foo", ""],
    })
    f = VowpalWabbitFeaturizer(inputCols=["num", "cat"],
                               stringSplitInputCols=["txt"], numBits=15)
    out = f.transform(df)
    vecs = out["features"].to_numpy()
    assert all(isinstance(v, SparseVector) for v in vecs)
    assert vecs[0].size == 1 << 15
    # row0: num + cat + 2 tokens = 4 features (assuming no collisions)
    assert len(vecs[0].indices) == 4
    # zero numeric dropped: row1 has cat + 1 token = 2
    assert len(vecs[1].indices) == 2
    # same string → same hash
    same_cat = set(vecs[0].indices) & set(vecs[2].indices)
    assert len(same_cat) >= 1


def test_featurizer_deterministic_and_sorted():
    df = pd.DataFrame({"a": ["x"], "b": [2.0]})
    f = VowpalWabbitFeaturizer(inputCols=["a", "b"], numBits=10)
    v1 = f.transform(df)["features"].iloc[0]
    v2 = f.transform(df)["features"].iloc[0]
    assert v1 == v2
    assert (np.diff(v1.indices) > 0).all()


def test_interactions():
    a = SparseVector(1 << 10, [1, 5], [1.0, 2.0])
    b = SparseVector(1 << 10, [3], [4.0])
    df = pd.DataFrame({"c1": [a], "c2": [b]})
    t = VowpalWabbitInteractions(inputCols=["c1", "c2"], numBits=10)
    v = t.transform(df)["interactions"].iloc[0]
    assert len(v.indices) == 2
    assert sorted(v.values.tolist()) == [4.0, 8.0]


def _hashed_binary_data(n=5000, nf=30, bits=16, seed=0):
    rng = np.random.default_rng(seed)
    size = 1 << bits
    feat_ids = rng.integers(0, size, size=200)
    w_true = rng.normal(size=size) * 0.0
    w_true[feat_ids] = rng.normal(size=len(feat_ids))
    rows = []
    labels = []
    for _ in range(n):
        k = rng.integers(5, nf)
        idx = np.unique(rng.choice(feat_ids, size=k))
        val = rng.normal(size=len(idx)).astype(np.float32)
        margin = float((w_true[idx] * val).sum())
        rows.append(SparseVector(size, idx.astype(np.int32), val))
        labels.append(1.0 if margin + rng.normal() * 0.3 > 0 else 0.0)
    return pd.DataFrame({"features": rows, "label": labels})


def test_vw_classifier_learns():
    from sklearn.metrics import roc_auc_score
    df = _hashed_binary_data()
    m = VowpalWabbitClassifier(numPasses=5, numBits=16, learningRate=0.5).fit(df)
    out = m.transform(df)
    prob = np.stack(out["probability"].to_numpy())[:, 1]
    assert roc_auc_score(df["label"], prob) > 0.85
    stats = m.getPerformanceStatistics()
    assert stats["numberOfExamplesPerPass"].iloc[0] == len(df)


def test_vw_regressor_learns():
    rng = np.random.default_rng(1)
    size = 1 << 14
    n = 4000
    w_true = rng.normal(size=size) * 0.05
    rows, ys = [], []
    for _ in range(n):
        idx = np.unique(rng.integers(0, size, size=20))
        val = np.ones(len(idx), dtype=np.float32)
        rows.append(SparseVector(size, idx.astype(np.int32), val))
        ys.append(float((w_true[idx] * val).sum()))
    df = pd.DataFrame({"features": rows, "label": ys})
    m = VowpalWabbitRegressor(numPasses=10, numBits=14, learningRate=0.3,
                              holdoutOff=True).fit(df)
    pred = m.transform(df)["prediction"].to_numpy()
    y = np.asarray(ys)
    ss_res = ((pred - y) ** 2).sum()
    ss_tot = ((y - y.mean()) ** 2).sum()
    assert 1 - ss_res / ss_tot > 0.5


def test_vw_args_string():
    est = VowpalWabbitRegressor(passThroughArgs="--l2 0.01 -b 20 --passes 3")
    est._parse_args()
    assert est.get("l2") == 0.01
    assert est.get("numBits") == 20
    assert est.get("numPasses") == 3


def test_vw_save_load(tmp_path):
    import os
    df = _hashed_binary_data(n=500)
    m = VowpalWabbitClassifier(numPasses=2, numBits=16).fit(df)
    p1 = np.stack(m.transform(df)["probability"].to_numpy())
    path = os.path.join(tmp_path, "vw")
    m.save(path)
    from mmlspark_amd.models.vw.estimators import VowpalWabbitClassificationModel
    m2 = VowpalWabbitClassificationModel.load(path)
    p2 = np.stack(m2.transform(df)["probability"].to_numpy())
    np.testing.assert_allclose(p1, p2)


def test_contextual_bandit():
    rng = np.random.default_rng(2)
    size = 1 << 14
    n = 2000
    n_actions = 3
    # context bit decides which action is cheapest
    rows = []
    for _ in range(n):
        ctx = int(rng.integers(0, n_actions))
        shared = SparseVector(size, [100 + ctx], [1.0])
        actions = [SparseVector(size, [2000 + a], [1.0])
                   for a in range(n_actions)]
        logged = int(rng.integers(0, n_actions))
        cost = 0.0 if logged == ctx else 1.0
        rows.append({"shared": shared, "features": actions,
                     "chosenAction": logged + 1, "cost": cost,
                     "probability": 1.0 / n_actions, "ctx": ctx})
    df = pd.DataFrame(rows)
    cb = VowpalWabbitContextualBandit(numPasses=5, numBits=14,
                                      learningRate=0.5).fit(df)
    out = cb.transform(df)
    picked = out["prediction"].to_numpy() - 1
    ctx = df["ctx"].to_numpy()
    assert (picked == ctx).mean() > 0.9


def test_vw_normalized_adaptive_flags():
    from sklearn.metrics import roc_auc_score
    df = _hashed_binary_data(n=3000)
    # scale one feature's values up 100x — normalized updates should cope
    for v in df["features"]:
        v.values[:1] *= 100.0
    m_norm = VowpalWabbitClassifier(numPasses=5, numBits=16,
                                    normalized=True).fit(df)
    p = np.stack(m_norm.transform(df)["probability"].to_numpy())[:, 1]
    assert roc_auc_score(df["label"], p) > 0.8
    # plain (non-adaptive) SGD path on well-scaled data
    df2 = _hashed_binary_data(n=3000, seed=3)
    m_plain = VowpalWabbitClassifier(numPasses=5, numBits=16, adaptive=False,
                                     learningRate=0.05).fit(df2)
    p2 = np.stack(m_plain.transform(df2)["probability"].to_numpy())[:, 1]
    assert roc_auc_score(df2["label"], p2) > 0.75
    # arg-string flags parse
    est = VowpalWabbitClassifier(passThroughArgs="--normalized --adaptive")
    est._parse_args()
    assert est.get("normalized") and est.get("adaptive")


def test_invariant_update_semantics():
    """--invariant (VowpalWabbitBase.scala arg surface): a weight-h update
    equals h sequential weight-1 updates as h*eta -> 0, and never overshoots
    the label for ANY h (closed-form gradient-flow integration)."""
    import torch
    from mmlspark_amd.models.vw import sgd_ref

    def one_example(h, invariant, loss="squared", n_upd=1, lr=0.5,
                    power_t=0.5):
        idx = torch.tensor([0, 1], dtype=torch.int32)
        val = torch.tensor([1.0, 2.0])
        off = torch.tensor([0, 2], dtype=torch.int64)
        y = torch.tensor([3.0 if loss == "squared" else 1.0])
        w = torch.zeros(8)
        g = torch.zeros(8)
        for _ in range(n_upd):
            sgd_ref.vw_sgd_minibatch(
                idx, val, off, y, w, g, lr, 0.0, power_t, loss,
                ex_weight=torch.tensor([float(h)]), invariant=invariant)
        pred = float(w[0] * 1.0 + w[1] * 2.0)
        return pred

    # huge importance weight: invariant converges exactly to the label;
    # plain AdaGrad self-normalizes to a fixed lr-sized step far from it
    p_inv = one_example(1e6, True)
    p_plain = one_example(1e6, False)
    assert abs(p_inv - 3.0) < 1e-3          # asymptote = label, no overshoot
    assert abs(p_plain - 3.0) > 1.0
    # invariance: with constant rates (power_t=0) one h=4 update is EXACTLY
    # 4 sequential h=1 updates (exp decay composes); AdaGrad breaks exactness
    p4 = one_example(4, True, lr=0.01, power_t=0.0)
    pseq = one_example(1, True, lr=0.01, n_upd=4, power_t=0.0)
    assert abs(p4 - pseq) < 1e-6
    # logistic: q = y*p strictly grows toward +inf but by a bounded step
    pl = one_example(50, True, loss="logistic", lr=0.1)
    assert 0 < pl < 20 and np.isfinite(pl)
    # hinge: lands exactly ON the margin (y*p == 1), never past
    ph = one_example(1e5, True, loss="hinge", lr=0.5)
    assert abs(ph - 1.0) < 1e-5


def test_invariant_flag_parses_and_trains():
    rng = np.random.default_rng(5)
    n = 4000
    X = rng.normal(size=(n, 6)).astype(np.float32)
    wstar = np.array([2, -1.5, 1, 0, 0, 0], dtype=np.float32)
    y = (X @ wstar + rng.normal(size=n) * 0.2 > 0).astype(np.float32)
    df = pd.DataFrame({"text": [
        " ".join(f"f{j}:{X[i, j]:.4f}" for j in range(6)) for i in range(n)],
        "label": y})
    feat = VowpalWabbitFeaturizer(inputCols=["text"], outputCol="features")
    fdf = feat.transform(df)
    m = VowpalWabbitClassifier(
        passThroughArgs="--invariant --adaptive --loss_function logistic",
        numPasses=8, learningRate=0.5).fit(fdf)
    out = m.transform(fdf)
    acc = float((out["prediction"].to_numpy() == y).mean())
    assert acc > 0.9


def test_interactions_estimator_level():
    """Estimator-level interactions (-q analog): crossing two informative
    namespaces lets a linear model learn an XOR-ish product signal."""
    rng = np.random.default_rng(7)
    n = 3000
    a = rng.choice([-1.0, 1.0], size=n).astype(np.float32)
    b = rng.choice([-1.0, 1.0], size=n).astype(np.float32)
    y = (a * b > 0).astype(np.float32)  # pure interaction, no main effects
    df = pd.DataFrame({"fa": list(np.eye(2, dtype=np.float32)[(a > 0).astype(int)]),
                       "fb": list(np.eye(2, dtype=np.float32)[(b > 0).astype(int)]),
                       "label": y})
    plain = VowpalWabbitClassifier(featuresCol="fa", additionalFeatures=["fb"],
                                   numPasses=6).fit(df)
    crossed = VowpalWabbitClassifier(featuresCol="fa", additionalFeatures=["fb"],
                                     interactions=["fa,fb"],
                                     numPasses=6).fit(df)
    acc_p = (plain.transform(df)["prediction"].to_numpy() == y).mean()
    acc_c = (crossed.transform(df)["prediction"].to_numpy() == y).mean()
    assert acc_c > 0.95
    assert acc_p < 0.7  # no main effects: plain linear model can't learn it


def test_featurizer_order_bits_and_prefix():
    df = pd.DataFrame({"colA": ["x"], "colB": ["x"]})
    f = VowpalWabbitFeaturizer(inputCols=["colA", "colB"], numBits=10,
                               preserveOrderNumBits=2)
    v = f.transform(df)["features"].iloc[0]
    # order prefix puts colA features in the bottom quarter, colB in the next
    assert all(i < 256 for i in v.indices[:1])
    assert any(256 <= i < 512 for i in v.indices)
    # prefix off: same token in two columns hashes differently only via seed
    f2 = VowpalWabbitFeaturizer(inputCols=["colA"],
                                prefixStringsWithColumnName=False, numBits=10)
    f3 = VowpalWabbitFeaturizer(inputCols=["colA"], numBits=10)
    i2 = f2.transform(df)["features"].iloc[0].indices[0]
    i3 = f3.transform(df)["features"].iloc[0].indices[0]
    assert i2 != i3  # "x" vs "colAx" under the same namespace seed


def test_args_alias_and_label_conversion():
    rng = np.random.default_rng(8)
    n = 2000
    X = rng.normal(size=(n, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = VowpalWabbitClassifier(args="--learning_rate 0.7 --passes 4").fit(df)
    assert m.get("numBits") == 18
    acc = (m.transform(df)["prediction"].to_numpy() == y).mean()
    assert acc > 0.9
    m2 = VowpalWabbitClassifier(labelConversion=False, numPasses=4).fit(df)
    assert (m2.transform(df)["prediction"].to_numpy() == y).mean() > 0.7


def test_bfgs_mode():
    """--bfgs (VowpalWabbitBase arg surface): full-batch L-BFGS over the
    hashed table converges to a better solution than 1-pass SGD on the
    same budget and reaches near-separable accuracy."""
    rng = np.random.default_rng(21)
    n = 3000
    X = rng.normal(size=(n, 8)).astype(np.float32)
    wstar = rng.normal(size=8).astype(np.float32)
    y = (X @ wstar > 0).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = VowpalWabbitClassifier(passThroughArgs="--bfgs",
                               lossFunction="logistic").fit(df)
    acc = (m.transform(df)["prediction"].to_numpy() == y).mean()
    assert acc > 0.97, acc
    # regressor path + l2
    yr = (X @ wstar).astype(np.float32)
    dfr = pd.DataFrame({"features": list(X), "label": yr})
    r = VowpalWabbitRegressor(bfgs=True, l2=1e-6).fit(dfr)
    pred = r.transform(dfr)["prediction"].to_numpy()
    assert np.corrcoef(pred, yr)[0, 1] > 0.99


def test_raw_quadratic_namespace_crossing():
    """-q on raw (unfeaturized) input (VowpalWabbitBase arg surface): a
    multiplicative task that a linear model cannot learn becomes learnable
    with '-q ab' namespace crossing."""
    from mmlspark_amd.models.vw.estimators import VowpalWabbitClassifier
    rng = np.random.default_rng(3)
    n = 6000
    xa = rng.choice([-1.0, 1.0], size=n).astype(np.float32)
    xb = rng.choice([-1.0, 1.0], size=n).astype(np.float32)
    y = (xa * xb > 0).astype(np.float32)  # XOR-like: linear AUC ≈ 0.5
    df = pd.DataFrame({
        "afeat": [np.array([v, 1.0], dtype=np.float32) for v in xa],
        "bfeat": [np.array([v, 1.0], dtype=np.float32) for v in xb],
        "label": y,
    })
    kw = dict(featuresCol="afeat", additionalFeatures=["bfeat"],
              numPasses=8, numBits=18, learningRate=0.5, adaptive=True,
              holdoutOff=True)
    m_lin = VowpalWabbitClassifier(**kw).fit(df)
    acc_lin = (m_lin.transform(df)["prediction"].to_numpy() == y).mean()
    m_q = VowpalWabbitClassifier(passThroughArgs="-q ab", **kw).fit(df)
    acc_q = (m_q.transform(df)["prediction"].to_numpy() == y).mean()
    assert acc_lin < 0.62, acc_lin
    assert acc_q > 0.95, acc_q
    # ':' wildcard resolves to all column pairs
    m_w = VowpalWabbitClassifier(passThroughArgs="--quadratic ::", **kw).fit(df)
    acc_w = (m_w.transform(df)["prediction"].to_numpy() == y).mean()
    assert acc_w > 0.95, acc_w


def test_multipass_holdout_early_terminate():
    """VW multi-pass semantics: every 10th example held out; training stops
    after earlyTerminate passes without holdout improvement (and
    --holdout_off disables all of it)."""
    from mmlspark_amd.models.vw.estimators import VowpalWabbitRegressor
    rng = np.random.default_rng(5)
    n, d = 2000, 32
    X = rng.normal(size=(n, d)).astype(np.float32)
    w = rng.normal(size=d)
    y = (X @ w).astype(np.float32)  # learnable fast → later passes plateau
    df = pd.DataFrame({"features": list(X), "label": y})
    m = VowpalWabbitRegressor(numPasses=40, learningRate=0.5, numBits=12,
                              adaptive=True, earlyTerminate=2).fit(df)
    stats = m.getPerformanceStatistics()
    ran = int(stats.iloc[0]["ipassCurrent"])
    assert ran < 40, ran  # early-terminated
    m2 = VowpalWabbitRegressor(numPasses=5, learningRate=0.5, numBits=12,
                               adaptive=True, holdoutOff=True).fit(df)
    assert int(m2.getPerformanceStatistics().iloc[0]["ipassCurrent"]) == 5
    # passThroughArgs spellings parse
    m3 = VowpalWabbitRegressor(
        passThroughArgs="--passes 3 --holdout_off", learningRate=0.5,
        numBits=12).fit(df)
    assert int(m3.getPerformanceStatistics().iloc[0]["ipassCurrent"]) == 3


def test_contextual_bandit_epsilon_pmf():
    """The CB model emits the epsilon-greedy action distribution
    (VW --epsilon pmf): greedy gets 1-eps+eps/K, the rest eps/K."""
    from mmlspark_amd.models.vw.estimators import VowpalWabbitContextualBandit
    rng = np.random.default_rng(0)
    size = 1 << 14
    rows = []
    for _ in range(300):
        shared = SparseVector(size, [100], [1.0])
        actions = [SparseVector(size, [2000 + a], [1.0]) for a in range(3)]
        chosen = int(rng.integers(1, 4))
        cost = float(chosen != 2)  # action 2 is best
        rows.append({"shared": shared, "features": actions,
                     "chosenAction": chosen, "cost": cost,
                     "probability": 1 / 3})
    df = pd.DataFrame(rows)
    m = VowpalWabbitContextualBandit(numPasses=4, numBits=14,
                                     learningRate=0.5, epsilon=0.3).fit(df)
    out = m.transform(df.head(10))
    assert "probabilities" in out.columns
    for _, r in out.iterrows():
        p = np.asarray(r["probabilities"])
        assert p.shape == (3,)
        np.testing.assert_allclose(p.sum(), 1.0, atol=1e-6)
        assert abs(p.max() - (0.7 + 0.1)) < 1e-6
        assert int(np.argmax(p)) + 1 == r["prediction"]


def test_contextual_bandit_additional_shared_features():
    """Extra shared columns change (and help) the policy; they are merged
    into the shared namespace with per-column murmur seeds."""
    rng = np.random.default_rng(4)
    size = 1 << 14
    rows = []
    for _ in range(1500):
        ctx = int(rng.integers(0, 3))
        # the discriminative context lives ONLY in the extra shared column
        extra = SparseVector(size, [300 + ctx], [1.0])
        shared = SparseVector(size, [100], [1.0])
        actions = [SparseVector(size, [2000 + a], [1.0]) for a in range(3)]
        logged = int(rng.integers(0, 3))
        cost = 0.0 if logged == ctx else 1.0
        rows.append({"shared": shared, "extraCtx": extra, "features": actions,
                     "chosenAction": logged + 1, "cost": cost,
                     "probability": 1 / 3, "ctx": ctx})
    df = pd.DataFrame(rows)
    cb = VowpalWabbitContextualBandit(
        numPasses=5, numBits=14, learningRate=0.5,
        additionalSharedFeatures=["extraCtx"]).fit(df)
    out = cb.transform(df)
    acc = ((out["prediction"].to_numpy() - 1) == df["ctx"].to_numpy()).mean()
    assert acc > 0.9
    # without the extra column the context is invisible → near-chance policy
    cb0 = VowpalWabbitContextualBandit(numPasses=5, numBits=14,
                                       learningRate=0.5).fit(df)
    out0 = cb0.transform(df)
    acc0 = ((out0["prediction"].to_numpy() - 1) == df["ctx"].to_numpy()).mean()
    assert acc0 < 0.6


def test_contextual_bandit_metrics_ips_snips():
    """IPS/SNIPS off-policy estimators: with uniform logging (p=1/K) and a
    deterministic target policy, IPS must be unbiased for the target
    policy's expected cost and SNIPS must match it up to normalization."""
    from mmlspark_amd.models.vw.estimators import ContextualBanditMetrics
    rng = np.random.default_rng(0)
    K = 4
    # true per-action costs; target policy always picks action 0 (cost 0.2)
    costs = np.array([0.2, 0.8, 0.5, 1.0])
    m = ContextualBanditMetrics()
    for _ in range(20000):
        logged = int(rng.integers(0, K))
        m.add(prob_logged=1.0 / K, cost=float(costs[logged]),
              prob_pred_matches=1.0 if logged == 0 else 0.0)
    assert abs(m.ips_estimate - costs[0]) < 0.03
    assert abs(m.snips_estimate - costs[0]) < 0.03
    assert m.total == 20000
