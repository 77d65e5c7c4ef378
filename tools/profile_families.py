"""cProfile sweep of family transform paths on GPU — hunting framework
overhead around fast kernels (the SAR/SHAP class of wins)."""
import cProfile, pstats, sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np, pandas as pd, torch

def top(pr, n=6, label=""):
    st = pstats.Stats(pr)
    print(f"--- {label}")
    for fn, (cc, nc, tt, ct, callers) in sorted(st.stats.items(), key=lambda kv: -kv[1][2])[:n]:
        f = f"{fn[0].split('/')[-1]}:{fn[1]}:{fn[2]}"
        print(f"  tot {tt:6.2f}s cum {ct:6.2f}s n={nc:7d}  {f[:70]}")

rng = np.random.default_rng(0)

# KNN
from mmlspark_amd.models.knn import KNN
idx_df = pd.DataFrame({"features": list(rng.normal(size=(200_000, 128)).astype(np.float32))})
q = pd.DataFrame({"features": list(rng.normal(size=(50_000, 128)).astype(np.float32))})
model = KNN(k=10).fit(idx_df)
model.transform(q.head(100))
pr = cProfile.Profile(); pr.enable()
t0=time.perf_counter(); model.transform(q); dt=time.perf_counter()-t0
pr.disable(); print(f"knn: {50000/dt:.0f} q/s"); top(pr, 6, "knn")

# iforest
from mmlspark_amd.models.iforest import IsolationForest
train = pd.DataFrame({"features": list(rng.normal(size=(20_000, 16)).astype(np.float32))})
m2 = IsolationForest(numEstimators=100).fit(train)
X = pd.DataFrame({"features": list(rng.normal(size=(2_000_000, 16)).astype(np.float32))})
m2.transform(X.head(100))
pr = cProfile.Profile(); pr.enable()
t0=time.perf_counter(); m2.transform(X); dt=time.perf_counter()-t0
pr.disable(); print(f"iforest: {2_000_000/dt/1e6:.2f} M rows/s"); top(pr, 6, "iforest")

# GBDT transform (batch scoring path itself)
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier
Xg = rng.normal(size=(2_000_000, 28)).astype(np.float32)
dfg = pd.DataFrame({"features": list(Xg[:20000]), "label": (Xg[:20000,0]>0).astype(np.float32)})
mg = LightGBMClassifier(numIterations=100, numLeaves=31, device="cuda" if torch.cuda.is_available() else "cpu").fit(dfg)
big = pd.DataFrame({"features": list(Xg)})
mg.transform(big.head(100))
pr = cProfile.Profile(); pr.enable()
t0=time.perf_counter(); mg.transform(big); dt=time.perf_counter()-t0
pr.disable(); print(f"gbdt transform: {2_000_000/dt/1e6:.2f} M rows/s"); top(pr, 6, "gbdt transform")
