"""ModelEquality — structural comparison of two saved stages
(core/.../core/utils/ModelEquality.scala parity; used by the fuzzing layer
to assert python↔saved correspondence)."""
from __future__ import annotations

import json
import os

import numpy as np


def assert_model_equality(path_a: str, path_b: str, rtol: float = 1e-6):
    with open(os.path.join(path_a, "metadata.json")) as f:
        ma = json.load(f)
    with open(os.path.join(path_b, "metadata.json")) as f:
        mb = json.load(f)
    assert ma["class"] == mb["class"], (ma["class"], mb["class"])
    assert ma["paramMap"] == mb["paramMap"], "paramMap mismatch"
    assert set(ma.get("complex", {})) == set(mb.get("complex", {}))
    for name, kind in ma.get("complex", {}).items():
        fa = os.path.join(path_a, f"data_{name}")
        fb = os.path.join(path_b, f"data_{name}")
        if kind == "ndarray":
            np.testing.assert_allclose(np.load(fa + ".npy"),
                                       np.load(fb + ".npy"), rtol=rtol)
        elif kind == "arrays":
            with np.load(fa + ".npz") as za, np.load(fb + ".npz") as zb:
                assert set(za.files) == set(zb.files)
                for k in za.files:
                    np.testing.assert_allclose(za[k], zb[k], rtol=rtol)
        elif kind == "stage":
            assert_model_equality(fa, fb, rtol)
        elif kind == "stages":
            with open(os.path.join(fa, "n.json")) as f:
                n = json.load(f)
            for i in range(n):
                assert_model_equality(os.path.join(fa, str(i)),
                                      os.path.join(fb, str(i)), rtol)
        else:
            ext = {"json": ".json", "bytes": ".bin", "dataframe": ".parquet",
                   "tensors": ".pt"}.get(kind, "")
            with open(fa + ext, "rb") as f1, open(fb + ext, "rb") as f2:
                assert f1.read() == f2.read(), f"complex param {name} differs"
