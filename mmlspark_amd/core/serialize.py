"""Pipeline-stage persistence: metadata JSON + typed side files.

Format (direct analog of the reference's ComplexParamsWriter/Reader,
org/apache/spark/ml/ComplexParamsSerializer.scala:35,147 — SparkML metadata
JSON next to ``data_<param>`` side files for params that are not JSON-able):

    <path>/metadata.json        {"class", "uid", "timestamp", "paramMap",
                                 "complex": {name: kind}}
    <path>/data_<param>.<ext>   side file per complex param

Side-file kinds:
    ndarray  -> .npy            (numpy)
    arrays   -> .npz            (dict of numpy arrays)
    dataframe-> .parquet        (pandas via pyarrow)
    stage    -> subdirectory    (nested PipelineStage, recursive)
    stages   -> subdirectory with n subdirs (list of stages)
    tensors  -> .pt             (torch state dict / tensor payload)
    bytes    -> .bin
    json     -> .json           (plain JSON-able but large)
"""
from __future__ import annotations

import json
import os
import time
from typing import Any, Dict

import numpy as np


def _is_jsonable(v) -> bool:
    try:
        json.dumps(v)
        return True
    except (TypeError, ValueError):
        return False


def _classify(v):
    import pandas as pd
    import torch
    from .pipeline import PipelineStage

    if isinstance(v, np.ndarray):
        return "ndarray"
    if isinstance(v, dict) and v and all(isinstance(x, np.ndarray) for x in v.values()):
        return "arrays"
    if isinstance(v, pd.DataFrame):
        return "dataframe"
    if isinstance(v, PipelineStage):
        return "stage"
    if isinstance(v, (list, tuple)) and v and all(isinstance(x, PipelineStage) for x in v):
        return "stages"
    if isinstance(v, torch.Tensor) or (
        isinstance(v, dict) and v and all(isinstance(x, torch.Tensor) for x in v.values())
    ):
        return "tensors"
    if isinstance(v, (bytes, bytearray)):
        return "bytes"
    return "json"


def save_stage(stage, path: str, overwrite: bool = True):
    import pandas as pd
    import torch

    os.makedirs(path, exist_ok=True)
    meta_path = os.path.join(path, "metadata.json")
    if os.path.exists(meta_path) and not overwrite:
        raise FileExistsError(path)

    param_map: Dict[str, Any] = {}
    complex_map: Dict[str, str] = {}
    for name, p in stage.params().items():
        if not stage.isSet(p) and p.default is None:
            continue
        v = stage.get(p)
        if v is None:
            continue
        if not p.is_complex and _is_jsonable(v):
            param_map[name] = v
            continue
        kind = _classify(v)
        complex_map[name] = kind
        base = os.path.join(path, f"data_{name}")
        if kind == "ndarray":
            np.save(base + ".npy", v)
        elif kind == "arrays":
            np.savez(base + ".npz", **v)
        elif kind == "dataframe":
            v.to_parquet(base + ".parquet")
        elif kind == "stage":
            save_stage(v, base)
        elif kind == "stages":
            os.makedirs(base, exist_ok=True)
            with open(os.path.join(base, "n.json"), "w") as f:
                json.dump(len(v), f)
            for i, s in enumerate(v):
                save_stage(s, os.path.join(base, str(i)))
        elif kind == "tensors":
            torch.save(v, base + ".pt")
        elif kind == "bytes":
            with open(base + ".bin", "wb") as f:
                f.write(bytes(v))
        else:  # json side file
            with open(base + ".json", "w") as f:
                json.dump(v, f)

    meta = {
        "class": f"{type(stage).__module__}.{type(stage).__name__}",
        "uid": stage.uid,
        "timestamp": int(time.time() * 1000),
        "framework": "mmlspark_amd",
        "paramMap": param_map,
        "complex": complex_map,
    }
    with open(meta_path, "w") as f:
        json.dump(meta, f, indent=1)


def load_stage(path: str):
    import pandas as pd
    import torch

    from .registry import lookup

    with open(os.path.join(path, "metadata.json")) as f:
        meta = json.load(f)
    cls = lookup(meta["class"])
    stage = cls.__new__(cls)
    # re-init Params plumbing without calling subclass __init__
    stage._paramMap = {}
    stage.uid = meta["uid"]
    extra = getattr(stage, "_post_deserialize_init", None)

    for name, v in meta["paramMap"].items():
        if stage.hasParam(name):
            stage.set(name, v)
    for name, kind in meta.get("complex", {}).items():
        base = os.path.join(path, f"data_{name}")
        if kind == "ndarray":
            v = np.load(base + ".npy", allow_pickle=False)
        elif kind == "arrays":
            with np.load(base + ".npz") as z:
                v = {k: z[k] for k in z.files}
        elif kind == "dataframe":
            v = pd.read_parquet(base + ".parquet")
        elif kind == "stage":
            v = load_stage(base)
        elif kind == "stages":
            with open(os.path.join(base, "n.json")) as f:
                n = json.load(f)
            v = [load_stage(os.path.join(base, str(i))) for i in range(n)]
        elif kind == "tensors":
            # state dicts / tensor payloads only — weights_only forbids
            # arbitrary pickle execution when pointed at untrusted paths.
            # (Set MMLSPARK_AMD_TRUSTED_LOAD=1 only for trusted legacy files
            # that predate the weights-only format.)
            if os.environ.get("MMLSPARK_AMD_TRUSTED_LOAD"):
                v = torch.load(base + ".pt", map_location="cpu",
                               weights_only=False)
            else:
                v = torch.load(base + ".pt", map_location="cpu",
                               weights_only=True)
        elif kind == "bytes":
            with open(base + ".bin", "rb") as f:
                v = f.read()
        else:
            with open(base + ".json") as f:
                v = json.load(f)
        stage.set(name, v)
    if extra is not None:
        extra()
    return stage
