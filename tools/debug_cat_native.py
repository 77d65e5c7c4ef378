"""Diagnose native-vs-Python grower divergence on categorical splits:
train both on identical GPU data and print the first differing tree/field."""
import os
import sys

import numpy as np
import pandas as pd
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from mmlspark_amd.models.gbdt.estimators import LightGBMClassifier  # noqa: E402

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
m_py = LightGBMClassifier(**kw).fit(df)
del os.environ["MMLSPARK_AMD_NO_NATIVE_GROWER"]

for ti, (tn, tp) in enumerate(zip(m_native.booster.trees,
                                  m_py.booster.trees)):
    for fld in ("feature", "thr_bin", "left", "right", "value", "count",
                "gain", "leaf_index", "cat_offset", "cat_words",
                "threshold"):
        a = getattr(tn, fld)
        b = getattr(tp, fld)
        if len(a) != len(b):
            print(f"tree {ti} {fld}: LEN native={len(a)} py={len(b)}")
            break
        eq = (a == b) | (np.isnan(a.astype(np.float64, copy=False))
                         & np.isnan(b.astype(np.float64, copy=False))) \
            if a.dtype.kind == "f" else (a == b)
        if not np.asarray(eq).all():
            bad = np.nonzero(~np.asarray(eq))[0]
            i = int(bad[0])
            print(f"tree {ti} {fld}: first diff at node {i}: "
                  f"native={a[i]!r} py={b[i]!r} ({len(bad)} diffs)")
            print("  native node:", {f2: getattr(tn, f2)[i] for f2 in
                                     ("feature", "thr_bin", "value", "count",
                                      "gain", "cat_offset")})
            print("  python node:", {f2: getattr(tp, f2)[i] for f2 in
                                     ("feature", "thr_bin", "value", "count",
                                      "gain", "cat_offset")})
            sys.exit(0)
print("identical")
