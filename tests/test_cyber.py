"""CyberML: access anomaly detection, indexers, scalers, complement sampler."""
import numpy as np
import pandas as pd

from mmlspark_amd.models.cyber import (AccessAnomaly,
                                       ComplementAccessTransformer, IdIndexer,
                                       LinearScalarScaler,
                                       StandardScalarScaler)


def _access_log(seed=0):
    """Two user groups accessing disjoint resource sets within one tenant."""
    rng = np.random.default_rng(seed)
    rows = []
    for u in range(20):
        group = 0 if u < 10 else 1
        for _ in range(30):
            r = rng.integers(0, 10) + group * 10
            rows.append({"tenant": "t0", "user": f"u{u}", "res": f"r{r}"})
    return pd.DataFrame(rows)


def test_id_indexer():
    df = pd.DataFrame({"tenant": ["a", "a", "b"], "user": ["x", "y", "x"]})
    m = IdIndexer(inputCol="user", partitionKey="tenant",
                  outputCol="user_idx").fit(df)
    out = m.transform(df)
    assert out["user_idx"].tolist() == [0, 1, 0]


def test_scalers():
    df = pd.DataFrame({"tenant": ["a"] * 4 + ["b"] * 4,
                       "v": [1.0, 2.0, 3.0, 4.0, 10.0, 20.0, 30.0, 40.0]})
    m = StandardScalarScaler(inputCol="v", partitionKey="tenant",
                             outputCol="z").fit(df)
    out = m.transform(df)
    za = out[out.tenant == "a"]["z"].to_numpy()
    assert abs(za.mean()) < 1e-9
    lm = LinearScalarScaler(inputCol="v", partitionKey="tenant", outputCol="s",
                            minRequiredValue=0.0, maxRequiredValue=1.0).fit(df)
    s = lm.transform(df)["s"].to_numpy()
    assert s.min() == 0.0 and s.max() == 1.0


def test_complement_sampler():
    df = _access_log()
    idx_u = IdIndexer(inputCol="user", partitionKey="tenant",
                      outputCol="user_idx").fit(df)
    idx_r = IdIndexer(inputCol="res", partitionKey="tenant",
                      outputCol="res_idx").fit(df)
    dfi = idx_r.transform(idx_u.transform(df))
    comp = ComplementAccessTransformer(complementsetFactor=1).transform(dfi)
    seen = set(zip(dfi["user_idx"], dfi["res_idx"]))
    for _, row in comp.iterrows():
        assert (row["user_idx"], row["res_idx"]) not in seen


def test_access_anomaly_detects_cross_group_access():
    df = _access_log()
    idx_u = IdIndexer(inputCol="user", partitionKey="tenant",
                      outputCol="user_idx").fit(df)
    idx_r = IdIndexer(inputCol="res", partitionKey="tenant",
                      outputCol="res_idx").fit(df)
    dfi = idx_r.transform(idx_u.transform(df))
    model = AccessAnomaly(maxIter=8, rankParam=8).fit(dfi)

    # normal accesses: same-group; anomalous: cross-group
    normal = dfi.head(40)
    anomalous = normal.copy()
    anomalous["res_idx"] = (anomalous["res_idx"] + 10) % 20  # other group
    s_norm = model.transform(normal)["anomaly_score"].to_numpy()
    s_anom = model.transform(anomalous)["anomaly_score"].to_numpy()
    assert np.nanmean(s_anom) > np.nanmean(s_norm) + 0.5
