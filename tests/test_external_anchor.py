"""External accuracy anchors.

The round-1 benchmark CSVs pinned this framework's own outputs — a
regression harness, not a parity proof.  These tests anchor against
artifacts this framework did NOT produce:

  1. a handcrafted LightGBM v3 native model text (the published format:
     ~leaf refs, decision_type bits, cat_boundaries/cat_threshold) whose
     predictions are computed BY HAND in the test;
  2. scikit-learn's HistGradientBoosting — an independent implementation
     of the same histogram-GBDT algorithm — matched on AUC/MSE within the
     reference CSVs' precision discipline
     (benchmarks_VerifyLightGBMClassifier.csv:2-5 pins AUC ±0.07).

(Stock LightGBM itself is not installable in this offline image; the
hand-verified format fixture + an independent-implementation quality
anchor cover the two things a stock-LightGBM fixture would prove.)
"""
import math
import os

import numpy as np
import pandas as pd
import pytest
import torch

from mmlspark_amd.models.gbdt.booster import Booster
from mmlspark_amd.models.gbdt.estimators import (LightGBMClassifier,
                                                 LightGBMRegressor)

GOLDEN = os.path.join(os.path.dirname(__file__), "resources", "golden",
                      "lightgbm_golden_v3.txt")


def _hand_predict(x):
    """Independent traversal of the golden model, straight from the
    LightGBM format spec: numeric decision_type=2 → x<=thr left;
    categorical decision_type=1 → bit int(x) of cat_threshold set → left."""
    # Tree 0: node0 f0<=0.5 → node1 (f1<=-1.25 → 0.2 else -0.3) else 0.55
    if x[0] <= 0.5:
        t0 = 0.2 if x[1] <= -1.25 else -0.3
    else:
        t0 = 0.55
    # Tree 1: node0 f2 in {1,4,5,9} (cat_threshold=562) → node1
    #         (f0<=1.75 → -0.15 else 0.25) else -0.4
    if int(x[2]) in (1, 4, 5, 9):
        t1 = -0.15 if x[0] <= 1.75 else 0.25
    else:
        t1 = -0.4
    return t0 + t1


def test_golden_lightgbm_text_import_exact():
    with open(GOLDEN) as f:
        b = Booster.load_from_string(f.read())
    assert b.objective == "binary"
    assert b.n_features == 3
    assert b.num_trees == 2
    assert b.feature_names == ["f0", "f1", "f2"]
    X = np.array([
        [0.0, -2.0, 1.0],   # L,L  cat-left,L  → 0.2 + (-0.15)
        [0.0,  0.0, 1.0],   # L,R             → -0.3 + (-0.15)
        [2.0,  0.0, 4.0],   # R    cat-left,R → 0.55 + 0.25
        [2.0,  0.0, 0.0],   # R    cat-right  → 0.55 + (-0.4)
        [0.5, -1.25, 9.0],  # boundary: <= goes LEFT on both
        [-3.0, 5.0, 7.0],
    ], dtype=np.float32)
    raw = b.predict_raw(torch.from_numpy(X)).squeeze(-1).numpy()
    expect = np.array([_hand_predict(x) for x in X], dtype=np.float64)
    np.testing.assert_allclose(raw, expect, atol=1e-6)
    # probability via the header's sigmoid
    prob = b.predict_prob(torch.from_numpy(X)).numpy()
    np.testing.assert_allclose(
        prob[:, 1], [1 / (1 + math.exp(-v)) for v in expect], atol=1e-6)


def test_golden_round_trip_through_exporter():
    """import → export → import is prediction-stable (format fidelity)."""
    with open(GOLDEN) as f:
        b = Booster.load_from_string(f.read())
    b2 = Booster.load_from_string(b.to_lightgbm_text())
    X = torch.from_numpy(
        np.random.default_rng(0).normal(size=(64, 3)).astype(np.float32))
    X[:, 2] = torch.randint(0, 10, (64,)).float()
    assert torch.allclose(b.predict_raw(X), b2.predict_raw(X), atol=1e-6)


def _make_binary(seed=7, n=8000, nf=20):
    rng = np.random.default_rng(seed)
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    logits = X @ w + 0.5 * np.sin(3 * X[:, 0]) + 0.5 * X[:, 1] * X[:, 2]
    y = (logits + rng.normal(size=n) > 0).astype(np.float32)
    return X, y


def test_auc_matches_sklearn_hist_gbdt():
    """Independent-implementation anchor: same data, same capacity — our
    AUC must be within the reference benchmark CSVs' tolerance class of
    sklearn's HistGradientBoostingClassifier (itself a LightGBM re-derivation)."""
    from sklearn.ensemble import HistGradientBoostingClassifier
    from sklearn.metrics import roc_auc_score
    X, y = _make_binary()
    Xtr, ytr = X[:6000], y[:6000]
    Xte, yte = X[6000:], y[6000:]

    df = pd.DataFrame({"features": list(Xtr), "label": ytr})
    ours = LightGBMClassifier(numIterations=100, numLeaves=31,
                              learningRate=0.1, minDataInLeaf=20).fit(df)
    dfe = pd.DataFrame({"features": list(Xte), "label": yte})
    p_ours = np.stack(ours.transform(dfe)["probability"].to_numpy())[:, 1]

    skl = HistGradientBoostingClassifier(max_iter=100, max_leaf_nodes=31,
                                         learning_rate=0.1,
                                         min_samples_leaf=20,
                                         early_stopping=False,
                                         random_state=0).fit(Xtr, ytr)
    p_skl = skl.predict_proba(Xte)[:, 1]

    a_ours = roc_auc_score(yte, p_ours)
    a_skl = roc_auc_score(yte, p_skl)
    # the reference pins AUC at precision 0.07 per dataset; we hold a
    # tighter 0.02 against the independent implementation and require
    # we are not behind it by more than 0.01
    assert abs(a_ours - a_skl) < 0.02, (a_ours, a_skl)
    assert a_ours > a_skl - 0.01, (a_ours, a_skl)
    assert a_ours > 0.9


def test_mse_matches_sklearn_hist_gbdt_regression():
    from sklearn.ensemble import HistGradientBoostingRegressor
    rng = np.random.default_rng(11)
    n, nf = 6000, 15
    X = rng.normal(size=(n, nf)).astype(np.float32)
    w = rng.normal(size=nf)
    y = (X @ w + 0.3 * X[:, 0] * X[:, 1]
         + 0.2 * rng.normal(size=n)).astype(np.float32)
    Xtr, ytr, Xte, yte = X[:4500], y[:4500], X[4500:], y[4500:]

    df = pd.DataFrame({"features": list(Xtr), "label": ytr})
    ours = LightGBMRegressor(numIterations=100, numLeaves=31,
                             learningRate=0.1, minDataInLeaf=20).fit(df)
    pred = ours.transform(pd.DataFrame({"features": list(Xte),
                                        "label": yte}))["prediction"]
    mse_ours = float(np.mean((pred.to_numpy() - yte) ** 2))

    skl = HistGradientBoostingRegressor(max_iter=100, max_leaf_nodes=31,
                                        learning_rate=0.1,
                                        min_samples_leaf=20,
                                        early_stopping=False,
                                        random_state=0).fit(Xtr, ytr)
    mse_skl = float(np.mean((skl.predict(Xte) - yte) ** 2))
    var = float(np.var(yte))
    # normalized MSE within 5% of the independent implementation
    assert mse_ours / var < mse_skl / var + 0.05, (mse_ours, mse_skl, var)
