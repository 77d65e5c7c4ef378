import time, numpy as np, pandas as pd, sys, os, cProfile, pstats
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from mmlspark_amd.explainers.shap import TabularSHAP
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
rng = np.random.default_rng(0)
n, m = 20000, 20
X = rng.normal(size=(n, m)).astype(np.float32)
cols = [f"f{i}" for i in range(m)]
df = pd.DataFrame(X, columns=cols)
df["label"] = (X[:,0] > 0).astype(np.float32)
model = LightGBMClassifier(featureCols=cols, numIterations=100, numLeaves=31, device="cuda").fit(df)
shap = TabularSHAP(inputCols=cols, model=model, targetCol="probability",
                   targetClasses=[1], numSamples=10000,
                   backgroundData=df.head(1000), rowBatch=16)
test = df.head(32)
shap.transform(test)
pr = cProfile.Profile(); pr.enable()
t0=time.perf_counter(); shap.transform(test); dt=time.perf_counter()-t0
pr.disable()
print(f"32 rows in {dt:.2f}s = {32/dt:.1f} expl/s")
st = pstats.Stats(pr)
for fn, (cc, nc, tt, ct, callers) in sorted(st.stats.items(), key=lambda kv: -kv[1][2])[:12]:
    f = f"{fn[0].split('/')[-1]}:{fn[1]}:{fn[2]}"
    print(f"  tot {tt:6.2f}s cum {ct:6.2f}s n={nc:6d}  {f[:75]}")
