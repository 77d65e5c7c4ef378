"""GBDT correctness on CPU: accuracy vs sklearn, objectives, modes, SHAP.

Accuracy-benchmark style regression tests in the spirit of the reference's
Benchmarks trait (core/.../core/test/benchmarks/Benchmarks.scala) with the
committed AUC-within-precision checks of
benchmarks_VerifyLightGBMClassifier.csv.
"""
import numpy as np
import pandas as pd
import pytest
import torch

from mmlspark_amd.models.gbdt.booster import Booster
from mmlspark_amd.models.gbdt.estimators import (
    LightGBMClassifier, LightGBMClassificationModel, LightGBMRanker,
    LightGBMRegressor)


def _auc(y, p):
    from sklearn.metrics import roc_auc_score
    return roc_auc_score(y, p)


def test_classifier_beats_bar(binary_df):
    m = LightGBMClassifier(numIterations=30, numLeaves=15, learningRate=0.2).fit(binary_df)
    out = m.transform(binary_df)
    prob = np.stack(out["probability"].to_numpy())[:, 1]
    y = binary_df["label"].to_numpy()
    assert _auc(y, prob) > 0.95


@pytest.mark.parametrize("boosting", ["gbdt", "goss", "dart", "rf"])
def test_boosting_types(binary_df, boosting):
    m = LightGBMClassifier(numIterations=15, numLeaves=15, learningRate=0.2,
                           boostingType=boosting, baggingFraction=0.8,
                           baggingFreq=1).fit(binary_df)
    out = m.transform(binary_df)
    prob = np.stack(out["probability"].to_numpy())[:, 1]
    y = binary_df["label"].to_numpy()
    bar = 0.8 if boosting == "rf" else 0.9
    assert _auc(y, prob) > bar, boosting


def test_classifier_matches_sklearn_quality(binary_df):
    from sklearn.ensemble import HistGradientBoostingClassifier
    X = np.stack(binary_df["features"].to_numpy())
    y = binary_df["label"].to_numpy()
    ours = LightGBMClassifier(numIterations=30, numLeaves=15, learningRate=0.2,
                              minDataInLeaf=20).fit(binary_df)
    p_ours = np.stack(ours.transform(binary_df)["probability"].to_numpy())[:, 1]
    sk = HistGradientBoostingClassifier(max_iter=30, learning_rate=0.2,
                                        max_leaf_nodes=15).fit(X, y)
    p_sk = sk.predict_proba(X)[:, 1]
    # same-family algorithm, same budget: within 0.02 AUC
    assert abs(_auc(y, p_ours) - _auc(y, p_sk)) < 0.02


def test_regressor(regression_df):
    m = LightGBMRegressor(numIterations=50, numLeaves=31).fit(regression_df)
    pred = m.transform(regression_df)["prediction"].to_numpy()
    y = regression_df["label"].to_numpy()
    assert np.sqrt(((pred - y) ** 2).mean()) < 0.5 * y.std()


@pytest.mark.parametrize("obj", ["regression_l1", "huber", "quantile", "poisson"])
def test_regression_objectives(regression_df, obj):
    df = regression_df.copy()
    if obj == "poisson":
        df["label"] = np.abs(df["label"].to_numpy()) + 0.1
    m = LightGBMRegressor(numIterations=30, numLeaves=15, objective=obj).fit(df)
    pred = m.transform(df)["prediction"].to_numpy()
    assert np.isfinite(pred).all()


def test_multiclass():
    rng = np.random.default_rng(3)
    n = 3000
    X = rng.normal(size=(n, 6)).astype(np.float32)
    y = np.digitize(X[:, 0] + X[:, 1], [-1.0, 1.0])  # 3 classes
    df = pd.DataFrame({"features": list(X), "label": y.astype(np.float32)})
    m = LightGBMClassifier(objective="multiclass", numIterations=15,
                           numLeaves=15).fit(df)
    out = m.transform(df)
    prob = np.stack(out["probability"].to_numpy())
    assert prob.shape[1] == 3
    assert np.allclose(prob.sum(axis=1), 1, atol=1e-5)
    acc = (out["prediction"].to_numpy() == y).mean()
    assert acc > 0.85


def test_ranker():
    rng = np.random.default_rng(4)
    n_q, per_q = 80, 20
    X = rng.normal(size=(n_q * per_q, 5)).astype(np.float32)
    rel = (X[:, 0] + 0.5 * X[:, 1] + rng.normal(size=len(X)) * 0.3)
    label = np.digitize(rel, np.quantile(rel, [0.5, 0.8, 0.95])).astype(np.float32)
    group = np.repeat(np.arange(n_q), per_q)
    df = pd.DataFrame({"features": list(X), "label": label, "group": group})
    m = LightGBMRanker(numIterations=20, numLeaves=15).fit(df)
    out = m.transform(df)
    # scores should correlate with relevance
    s = out["prediction"].to_numpy()
    lab = out["label"].to_numpy()
    assert np.corrcoef(s, lab)[0, 1] > 0.5


def test_shap_additivity(binary_df):
    m = LightGBMClassifier(numIterations=10, numLeaves=7).fit(binary_df)
    m.set("featuresShapCol", "shap")
    out = m.transform(binary_df.head(50))
    shap = np.stack(out["shap"].to_numpy())
    raw = np.stack(out["rawPrediction"].to_numpy())[:, 1]
    assert shap.shape[1] == 10 + 1
    assert np.abs(shap.sum(axis=1) - raw).max() < 1e-4


def test_native_model_string_roundtrip(binary_df):
    m = LightGBMClassifier(numIterations=8, numLeaves=7).fit(binary_df)
    s = m.getNativeModel()
    b = Booster.load_from_string(s)
    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy()))
    p1 = m.booster.predict_raw(X)
    p2 = b.predict_raw(X)
    assert torch.allclose(p1, p2)


def test_warm_start_model_string(binary_df):
    m1 = LightGBMClassifier(numIterations=5, numLeaves=7).fit(binary_df)
    m2 = LightGBMClassifier(numIterations=5, numLeaves=7,
                            modelString=m1.getNativeModel()).fit(binary_df)
    assert m2.booster.num_trees == 10


def test_num_batches(binary_df):
    m = LightGBMClassifier(numIterations=5, numLeaves=7, numBatches=2).fit(binary_df)
    assert m.booster.num_trees == 10  # 5 per batch


def test_early_stopping(binary_df):
    df = binary_df.copy()
    rng = np.random.default_rng(9)
    df["isVal"] = rng.random(len(df)) < 0.3
    m = LightGBMClassifier(numIterations=200, numLeaves=31, learningRate=0.5,
                           validationIndicatorCol="isVal",
                           earlyStoppingRound=5).fit(df)
    assert m.booster.num_trees < 200


def test_feature_importances(binary_df):
    m = LightGBMClassifier(numIterations=10, numLeaves=7).fit(binary_df)
    imp = m.getFeatureImportances("split")
    assert len(imp) == 10 and sum(imp) > 0
    impg = m.getFeatureImportances("gain")
    assert max(impg) > 0


def test_feature_cols_mode(binary_df):
    X = np.stack(binary_df["features"].to_numpy())
    df = pd.DataFrame({f"c{i}": X[:, i] for i in range(X.shape[1])})
    df["label"] = binary_df["label"].to_numpy()
    cols = [f"c{i}" for i in range(X.shape[1])]
    m = LightGBMClassifier(numIterations=10, numLeaves=7, featureCols=cols).fit(df)
    out = m.transform(df)
    assert _auc(df["label"], np.stack(out["probability"].to_numpy())[:, 1]) > 0.9


def test_empty_and_tiny_partition_robustness():
    """Analog of the reference's empty-partition tests
    (VerifyLightGBMClassifier.scala:594-643)."""
    rng = np.random.default_rng(5)
    X = rng.normal(size=(30, 4)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = LightGBMClassifier(numIterations=5, numLeaves=4, minDataInLeaf=1).fit(df)
    out = m.transform(df.head(0))
    assert len(out) == 0


def test_categorical_features():
    """categoricalSlotIndexes parity: the category id is the bin; splits are
    sorted one-vs-rest sets. A pure-categorical signal needs set splits."""
    rng = np.random.default_rng(11)
    n = 4000
    cat = rng.integers(0, 12, size=n).astype(np.float32)
    noise = rng.normal(size=(n, 3)).astype(np.float32)
    # non-monotone category → label mapping (numeric threshold can't separate)
    good = {1, 4, 7, 10}
    y = np.array([1.0 if int(c) in good else 0.0 for c in cat],
                 dtype=np.float32)
    X = np.column_stack([cat, noise])
    df = pd.DataFrame({"features": list(X.astype(np.float32)), "label": y})
    m = LightGBMClassifier(numIterations=10, numLeaves=15,
                           categoricalSlotIndexes=[0],
                           minDataInLeaf=5).fit(df)
    out = m.transform(df)
    acc = (out["prediction"].to_numpy() == y).mean()
    assert acc > 0.98, acc
    # trees actually used categorical splits
    assert any((t.cat_offset >= 0).any() for t in m.booster.trees)
    # save/load preserves bitsets
    b2 = m.booster.load_from_string(m.booster.save_to_string())
    X_t = torch.from_numpy(X.astype(np.float32))
    assert torch.allclose(m.booster.predict_raw(X_t), b2.predict_raw(X_t))


def test_init_score_col():
    rng = np.random.default_rng(12)
    n = 1000
    X = rng.normal(size=(n, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y,
                       "init": np.full(n, 4.0, dtype=np.float32)})
    m = LightGBMClassifier(numIterations=5, numLeaves=4, learningRate=0.5,
                           initScoreCol="init").fit(df)
    raw = m.booster.predict_raw(torch.from_numpy(X))
    m0 = LightGBMClassifier(numIterations=5, numLeaves=4, learningRate=0.5).fit(
        df.drop(columns=["init"]))
    raw0 = m0.booster.predict_raw(torch.from_numpy(X))
    # LightGBM semantics: booster output EXCLUDES the init margin, so the
    # trees must have learned to compensate the +4 offset downward
    assert float(raw.mean()) < float(raw0.mean()) - 0.5


def test_lightgbm_text_export(binary_df):
    m = LightGBMClassifier(numIterations=5, numLeaves=7).fit(binary_df)
    txt = m.booster.to_lightgbm_text()
    assert txt.startswith("tree\nversion=v3")
    assert txt.count("Tree=") == 5
    assert "end of trees" in txt
    assert f"max_feature_idx={10 - 1}" in txt
    # every tree block carries the required arrays
    for key in ("split_feature=", "threshold=", "left_child=", "leaf_value="):
        assert txt.count(key) == 5


def test_lightgbm_text_roundtrip(binary_df):
    """to_lightgbm_text → load_from_string gives identical predictions (the
    setModelString interop path: stock LightGBM model text loads here)."""
    from mmlspark_amd.models.gbdt.booster import Booster
    m = LightGBMClassifier(numIterations=8, numLeaves=15).fit(binary_df)
    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy()[:200]))
    txt = m.booster.to_lightgbm_text()
    b2 = Booster.load_from_string(txt)
    assert b2.n_features == m.booster.n_features
    assert b2.num_trees == m.booster.num_trees
    np.testing.assert_allclose(b2.predict_raw(X).numpy(),
                               m.booster.predict_raw(X).numpy(),
                               rtol=1e-5, atol=1e-5)


def test_lightgbm_text_roundtrip_multiclass():
    rng = np.random.default_rng(3)
    X = rng.normal(size=(1500, 6)).astype(np.float32)
    y = np.digitize(X[:, 0] + 0.5 * X[:, 1], [-0.5, 0.5]).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    from mmlspark_amd.models.gbdt.booster import Booster
    m = LightGBMClassifier(numIterations=5, numLeaves=7,
                           objective="multiclass").fit(df)
    txt = m.booster.to_lightgbm_text()
    b2 = Booster.load_from_string(txt)
    assert b2.n_outputs == 3
    Xt = torch.from_numpy(X[:100])
    np.testing.assert_allclose(b2.predict_raw(Xt).numpy(),
                               m.booster.predict_raw(Xt).numpy(),
                               rtol=1e-5, atol=1e-5)


def test_model_string_accepts_stock_lightgbm_text(binary_df):
    """A hand-written stock-LightGBM v3 model string (the reference's
    setModelString input format) scores through the full estimator path."""
    txt = "\n".join([
        "tree", "version=v3", "num_class=1", "num_tree_per_iteration=1",
        "label_index=0", "max_feature_idx=9",
        "objective=binary sigmoid:1",
        "feature_names=" + " ".join(f"f{i}" for i in range(10)),
        "feature_infos=" + " ".join(["none"] * 10), "",
        "Tree=0", "num_leaves=3", "num_cat=0",
        "split_feature=0 1", "threshold=0.25 -0.5", "decision_type=2 2",
        "left_child=1 -1", "right_child=-3 -2",
        "leaf_value=-1.5 0.5 2.0", "leaf_count=40 30 30",
        "internal_value=0 -0.4", "internal_count=100 70",
        "shrinkage=0.1", "",
        "end of trees", ""])
    m = LightGBMClassificationModel.loadNativeModelFromString(txt)
    X = np.array([[0.0, -1.0] + [0.0] * 8,    # f0<=.25, f1<=-.5 → leaf0 -1.5
                  [0.0, 0.0] + [0.0] * 8,     # f0<=.25, f1>-.5  → leaf1 0.5
                  [1.0, 0.0] + [0.0] * 8],    # f0>.25           → leaf2 2.0
                 dtype=np.float32)
    df = pd.DataFrame({"features": list(X)})
    out = m.transform(df)
    raw = np.array([v[1] for v in out["rawPrediction"]])
    np.testing.assert_allclose(raw, [-1.5, 0.5, 2.0], atol=1e-6)
    assert out["prediction"].tolist() == [0.0, 1.0, 1.0]


def test_early_stopping_predicts_with_best_iteration(binary_df):
    df = binary_df.copy()
    rng = np.random.default_rng(9)
    df["isVal"] = rng.random(len(df)) < 0.3
    m = LightGBMClassifier(numIterations=60, numLeaves=31, learningRate=0.8,
                           validationIndicatorCol="isVal",
                           earlyStoppingRound=3).fit(df)
    b = m.booster
    assert b.best_iteration >= 0
    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy()[:50]))
    default = b.predict_raw(X)
    best = b.predict_raw(X, num_iteration=b.best_iteration + 1)
    allt = b.predict_raw(X, num_iteration=b.num_trees)
    assert torch.allclose(default, best)
    assert not torch.allclose(default, allt)
    # persists through native-model text
    b2 = b.load_from_string(b.save_to_string())
    assert b2.best_iteration == b.best_iteration


def test_multiclass_shap_contrib_layout():
    rng = np.random.default_rng(3)
    n = 1500
    X = rng.normal(size=(n, 6)).astype(np.float32)
    y = np.digitize(X[:, 0] + X[:, 1], [-1.0, 1.0]).astype(np.float32)
    df = pd.DataFrame({"features": list(X), "label": y})
    m = LightGBMClassifier(objective="multiclass", numIterations=8,
                           numLeaves=7).fit(df)
    contrib = m.booster.predict_contrib(torch.from_numpy(X[:20]))
    K, nf = 3, 6
    assert contrib.shape == (20, K * (nf + 1))
    per = contrib.reshape(20, K, nf + 1)
    raw = m.booster.predict_raw(torch.from_numpy(X[:20])).numpy()
    # per-class additivity: sum of class contributions == class raw score
    np.testing.assert_allclose(per.sum(axis=2), raw, atol=1e-3)


def test_custom_fobj(binary_df):
    """fobj custom objective (FObjTrait analog): hand-written logistic
    grad/hess trains as well as the built-in binary objective."""
    def logistic_fobj(preds, label, weight):
        z = torch.sigmoid(preds.squeeze(-1))
        g = (z - label.float()).unsqueeze(-1)
        h = (z * (1 - z)).clamp_min(1e-16).unsqueeze(-1)
        return g, h

    m = LightGBMClassifier(numIterations=20, numLeaves=15, fobj=logistic_fobj,
                           objective="binary").fit(binary_df)
    prob = np.stack(m.transform(binary_df)["probability"].to_numpy())[:, 1]
    assert _auc(binary_df["label"].to_numpy(), prob) > 0.93


def test_is_unbalance_reweights(binary_df):
    """isUnbalance raises recall on the rare class via n_neg/n_pos weights."""
    df = binary_df.copy()
    pos = df[df.label == 1].head(40)  # make positives rare
    df = pd.concat([df[df.label == 0], pos]).reset_index(drop=True)
    base = LightGBMClassifier(numIterations=20, numLeaves=7).fit(df)
    bal = LightGBMClassifier(numIterations=20, numLeaves=7,
                             isUnbalance=True).fit(df)
    pb = np.stack(base.transform(df)["probability"].to_numpy())[:, 1]
    pw = np.stack(bal.transform(df)["probability"].to_numpy())[:, 1]
    y = df["label"].to_numpy()
    # weighted model shifts scores up on true positives
    assert pw[y == 1].mean() > pb[y == 1].mean()


def test_boost_from_average_flag(binary_df):
    m0 = LightGBMClassifier(numIterations=1, numLeaves=3, learningRate=0.0,
                            boostFromAverage=False).fit(binary_df)
    raw = m0.booster.predict_raw(
        torch.from_numpy(np.stack(binary_df["features"].to_numpy()[:4])))
    np.testing.assert_allclose(raw.numpy(), 0.0, atol=1e-6)  # zero init, lr=0


def test_start_iteration_predict_window(binary_df):
    m = LightGBMClassifier(numIterations=10, numLeaves=7).fit(binary_df)
    model = m
    X = torch.from_numpy(np.stack(binary_df["features"].to_numpy()[:50]))
    full = m.booster.predict_raw(X, 0, -1)
    tail = m.booster.predict_raw(X, 5, -1)   # trees 5..9 only
    head = m.booster.predict_raw(X, 0, 5)    # trees 0..4 only
    base = torch.from_numpy(m.booster.base_score)
    np.testing.assert_allclose((head + tail - base).numpy(), full.numpy(),
                               rtol=1e-4, atol=1e-4)
    model.set("startIteration", 5)
    out = model.transform(binary_df.head(50))
    raw_col = np.stack(out["rawPrediction"].to_numpy())[:, 1]
    np.testing.assert_allclose(raw_col, tail.squeeze(-1).numpy(), rtol=1e-4,
                               atol=1e-4)


def test_max_bin_by_feature_and_pos_bagging(binary_df):
    m = LightGBMClassifier(numIterations=10, numLeaves=7,
                           maxBinByFeature=[4] * 10,
                           posBaggingFraction=0.8, negBaggingFraction=0.5,
                           baggingFreq=1).fit(binary_df)
    prob = np.stack(m.transform(binary_df)["probability"].to_numpy())[:, 1]
    assert _auc(binary_df["label"].to_numpy(), prob) > 0.8  # still learns


def test_warm_start_from_booster_object(binary_df):
    m1 = LightGBMClassifier(numIterations=5, numLeaves=7).fit(binary_df)
    m2 = LightGBMClassifier(numIterations=5, numLeaves=7,
                            lightGBMBooster=m1.booster).fit(binary_df)
    assert m2.booster.num_trees == 10  # 5 warm + 5 new


def test_metric_direction_table():
    """direction keyed off the metric compared, not the objective."""
    from mmlspark_amd.models.gbdt.metrics import metric_higher_is_better
    assert metric_higher_is_better("auc")
    assert metric_higher_is_better("ndcg@5")
    assert metric_higher_is_better("map@10")
    assert not metric_higher_is_better("binary_logloss")
    assert not metric_higher_is_better("l2")
    assert not metric_higher_is_better("rmse")
    # unknown metric falls back to the provided default
    assert metric_higher_is_better("my_custom", True)
    assert not metric_higher_is_better("my_custom", False)


def test_early_stopping_auc_tracks_best_not_worst(binary_df):
    """With metric='auc' (higher-better) best_iteration must be the argmax
    of AUC over the eval history — the round-1 bug tracked the argmin."""
    df = binary_df.copy()
    rng = np.random.default_rng(5)
    df["isVal"] = rng.random(len(df)) < 0.3
    m = LightGBMClassifier(numIterations=60, numLeaves=31, learningRate=0.4,
                           metric="auc", validationIndicatorCol="isVal",
                           earlyStoppingRound=5).fit(df)
    evals = m._training_stats.evals
    aucs = [e["valid_0"]["auc"] for e in evals]
    bi = m.booster.best_iteration
    assert bi >= 0
    assert aucs[bi] == max(aucs), (bi, aucs)


def test_lightgbm_text_export_preserves_sigmoid(binary_df):
    m = LightGBMClassifier(numIterations=3, numLeaves=7).fit(binary_df)
    b = m.booster
    b.sigmoid = 2.5
    txt = b.to_lightgbm_text()
    assert "objective=binary sigmoid:2.5" in txt
    # round trip through our own text importer keeps the sigmoid
    b2 = Booster.load_from_string(txt)
    assert abs(float(b2.sigmoid) - 2.5) < 1e-6


def test_ranker_validation_ndcg_and_early_stopping():
    """Ranker validation now reports group-aware NDCG@k (evalAt) and early
    stopping treats it as higher-is-better (the round-1 ADVICE inversion)."""
    rng = np.random.default_rng(21)
    n_groups, per = 120, 20
    rows = []
    for gid in range(n_groups):
        q = rng.normal(size=4)
        for _ in range(per):
            x = rng.normal(size=4)
            rel = float(np.clip(round(2 + 1.5 * (q @ x) / 4
                                      + rng.normal() * 0.3), 0, 4))
            rows.append({"features": np.concatenate([q, x]).astype(np.float32),
                         "label": rel, "group": gid})
    df = pd.DataFrame(rows)
    df["isVal"] = (df["group"] % 5 == 0)
    m = LightGBMRanker(numIterations=60, numLeaves=15, learningRate=0.1,
                       groupCol="group", evalAt=[3, 5],
                       validationIndicatorCol="isVal",
                       earlyStoppingRound=8).fit(df)
    evals = m._training_stats.evals
    assert evals, "validation metrics must be recorded"
    keys = set(evals[0]["valid_0"].keys())
    assert keys == {"ndcg@3", "ndcg@5"}, keys
    ndcgs = [e["valid_0"]["ndcg@3"] for e in evals]
    bi = m.booster.best_iteration
    assert bi >= 0
    assert ndcgs[bi] == max(ndcgs), (bi, ndcgs)  # higher-is-better tracked
    assert max(ndcgs) > 0.6


def test_is_provide_training_metric(binary_df):
    """isProvideTrainingMetric logs per-iteration TRAIN metrics
    (TrainUtils.scala:117-128 parity) even without a validation split."""
    m = LightGBMClassifier(numIterations=5, numLeaves=7,
                           isProvideTrainingMetric=True).fit(binary_df)
    evals = m._training_stats.evals
    assert len(evals) == 5
    assert "training" in evals[0]
    assert "binary_logloss" in evals[0]["training"]
    losses = [e["training"]["binary_logloss"] for e in evals]
    assert losses[-1] < losses[0]  # training loss decreases


def test_bagging_seed_controls_sampling(binary_df):
    """baggingSeed must change bagging draws (same seed → same model)."""
    kw = dict(numIterations=8, numLeaves=15, baggingFraction=0.6,
              baggingFreq=1, seed=0)
    a = LightGBMClassifier(baggingSeed=1, **kw).fit(binary_df)
    b = LightGBMClassifier(baggingSeed=1, **kw).fit(binary_df)
    c = LightGBMClassifier(baggingSeed=2, **kw).fit(binary_df)
    assert a.booster.save_to_string() == b.booster.save_to_string()
    assert a.booster.save_to_string() != c.booster.save_to_string()


def test_leaf_output_zero_hessian_guard():
    """H + lambda_l2 == 0 (quantized hessians rounded to zero in a tiny
    leaf) must yield output 0.0, not ZeroDivisionError."""
    from mmlspark_amd.models.gbdt.trainer import TrainConfig, _leaf_output
    cfg = TrainConfig(lambda_l1=0.0, lambda_l2=0.0)
    assert _leaf_output(1.5, 0.0, cfg) == 0.0
    assert _leaf_output(-1.5, 0.0, cfg) == 0.0
    assert _leaf_output(1.5, 2.0, cfg) == -0.75


def test_booster_merge_appends_trees():
    """merge (mergeBooster:252) appends trees: the merged model's raw score
    equals the sum of both models' raws (additive forests, same objective)."""
    rng = np.random.default_rng(0)
    X = rng.normal(size=(400, 5)).astype(np.float32)
    y = (X[:, 0] > 0).astype(np.float64)
    df = pd.DataFrame({"features": list(X), "label": y})
    m1 = LightGBMClassifier(numIterations=3, numLeaves=7, seed=1).fit(df)
    m2 = LightGBMClassifier(numIterations=2, numLeaves=7, seed=2).fit(df)
    b1, b2 = m1.booster, m2.booster
    Xt = torch.from_numpy(X)
    r1 = b1.predict_raw(Xt).clone()
    r2 = b2.predict_raw(Xt).clone()
    import copy
    merged = copy.deepcopy(b1).merge(b2)
    assert merged.num_trees == b1.num_trees + b2.num_trees
    # merged keeps SELF's base_score and appends the other's trees, so the
    # expected raw is r1 + (r2 - base2) — the numBatches continuation
    # semantics (batch-2 boosters carry their lift in trees, not base)
    np.testing.assert_allclose(
        merged.predict_raw(Xt).numpy(),
        (r1 + r2).numpy() - b2.base_score, rtol=1e-5, atol=1e-5)
    # merged model round-trips through the JSON format
    from mmlspark_amd.models.gbdt.booster import Booster
    back = Booster.load_from_string(merged.save_to_string())
    np.testing.assert_allclose(back.predict_raw(Xt).numpy(),
                               merged.predict_raw(Xt).numpy())


def test_nan_features_train_and_predict():
    """NaN feature values must train (binned to the top bin) and score
    (routed like the top bin: raw <= thr is false) without poisoning
    outputs — the reference's useMissing/NaN tolerance."""
    rng = np.random.default_rng(0)
    X = rng.normal(size=(2000, 6)).astype(np.float32)
    y = (X[:, 0] + X[:, 1] > 0).astype(np.float64)
    Xn = X.copy()
    mask = rng.random(X.shape) < 0.1
    mask[:, 0] = False  # keep the signal feature mostly intact
    Xn[mask] = np.nan
    df = pd.DataFrame({"features": list(Xn), "label": y})
    m = LightGBMClassifier(numIterations=20, numLeaves=15).fit(df)
    out = m.transform(df)
    prob = np.stack(out["probability"].to_numpy())
    assert np.isfinite(prob).all()
    acc = (out["prediction"].to_numpy() == y).mean()
    assert acc > 0.85, acc
    # NaN routes exactly like +inf at predict time (both fail raw <= thr)
    row_nan = Xn[:1].copy(); row_nan[0, 2] = np.nan
    row_inf = Xn[:1].copy(); row_inf[0, 2] = np.inf
    p1 = m.booster.predict_raw(torch.from_numpy(np.nan_to_num(row_nan, nan=np.nan)))
    p2 = m.booster.predict_raw(torch.from_numpy(row_inf))
    np.testing.assert_allclose(p1.numpy(), p2.numpy())
