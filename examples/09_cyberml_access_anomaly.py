"""CyberML: detect anomalous user→resource access with collaborative
filtering (cyber/anomaly/collaborative_filtering.py AccessAnomaly parity).
Users normally touch their own team's resources; cross-team access should
score as anomalous."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import pandas as pd

from mmlspark_amd.models.cyber import AccessAnomaly, IdIndexer

rng = np.random.default_rng(0)

# two teams; members access their team's resources with high frequency
rows = []
for team, (users, resources) in enumerate(
        [([f"u{i}" for i in range(20)], [f"srv-a{i}" for i in range(15)]),
         ([f"v{i}" for i in range(20)], [f"srv-b{i}" for i in range(15)])]):
    for _ in range(3000):
        rows.append({"tenant": "acme",
                     "user": str(rng.choice(users)),
                     "res": str(rng.choice(resources))})
df = pd.DataFrame(rows)

idx_u = IdIndexer(inputCol="user", partitionKey="tenant",
                  outputCol="user_idx").fit(df)
idx_r = IdIndexer(inputCol="res", partitionKey="tenant",
                  outputCol="res_idx").fit(df)
index = lambda d: idx_r.transform(idx_u.transform(d))  # noqa: E731
dfi = index(df)

model = AccessAnomaly(maxIter=10, rankParam=8).fit(dfi)

# score normal (own-team) vs anomalous (cross-team) accesses
probe = pd.DataFrame({
    "tenant": ["acme"] * 4,
    "user": ["u0", "u1", "u0", "v0"],
    "res": ["srv-a0", "srv-a5", "srv-b3", "srv-a2"],   # last two cross-team
})
scored = model.transform(index(probe))
for _, r in scored.iterrows():
    print(f"{r.user:>4} -> {r.res:<8} anomaly={r.anomaly_score:7.3f}")

normal = scored.anomaly_score[:2].mean()
cross = scored.anomaly_score[2:].mean()
print(f"\nmean normal={normal:.3f}  mean cross-team={cross:.3f}")
assert cross > normal, "cross-team access should look more anomalous"
