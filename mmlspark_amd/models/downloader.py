"""ModelDownloader — local model repository with hash-verified schemas
(deep-learning/.../downloader/ModelDownloader.scala:26-90 parity; offline,
so the 'remote' repo is any mounted directory of state dicts)."""
from __future__ import annotations

import hashlib
import json
import os
from dataclasses import asdict, dataclass
from typing import List, Optional

import torch


@dataclass
class ModelSchema:
    name: str
    dataset: str
    modelType: str
    uri: str                 # path to the state-dict file
    hash: str                # sha256 of the file
    size: int
    inputNode: int = 0
    numLayers: int = 0
    layerNames: Optional[List[str]] = None


class ModelDownloader:
    def __init__(self, local_path: str):
        self.local_path = local_path
        os.makedirs(local_path, exist_ok=True)

    def _index_path(self):
        return os.path.join(self.local_path, "models.json")

    def list_models(self) -> List[ModelSchema]:
        if not os.path.exists(self._index_path()):
            return []
        with open(self._index_path()) as f:
            return [ModelSchema(**d) for d in json.load(f)]

    @staticmethod
    def _sha256(path: str) -> str:
        h = hashlib.sha256()
        with open(path, "rb") as f:
            for chunk in iter(lambda: f.read(1 << 20), b""):
                h.update(chunk)
        return h.hexdigest()

    def publish(self, name: str, module: torch.nn.Module, dataset: str = "",
                model_type: str = "torch", layer_names=None) -> ModelSchema:
        uri = os.path.join(self.local_path, f"{name}.pt")
        torch.save(module.state_dict(), uri)
        schema = ModelSchema(
            name=name, dataset=dataset, modelType=model_type, uri=uri,
            hash=self._sha256(uri), size=os.path.getsize(uri),
            layerNames=layer_names)
        models = [m for m in self.list_models() if m.name != name]
        models.append(schema)
        with open(self._index_path(), "w") as f:
            json.dump([asdict(m) for m in models], f, indent=1)
        return schema

    def download_by_name(self, name: str) -> ModelSchema:
        """'Download' = locate + hash-verify (remote repos are out of scope
        offline; the verification contract matches the reference)."""
        for m in self.list_models():
            if m.name == name:
                if self._sha256(m.uri) != m.hash:
                    raise IOError(f"hash mismatch for model {name}")
                return m
        raise KeyError(f"model {name!r} not in repo {self.local_path}")

    def load_state(self, name: str):
        return torch.load(self.download_by_name(name).uri, map_location="cpu")
