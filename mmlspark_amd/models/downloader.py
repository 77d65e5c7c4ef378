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
        return torch.load(self.download_by_name(name).uri, map_location="cpu",
                          weights_only=True)


class RemoteRepo:
    """HTTP model repository (HDFSRepo analog,
    downloader/ModelDownloader.scala:42): a base URL serving models.json
    plus the model payloads.  download_by_name fetches into a local
    ModelDownloader cache with sha256 verification — the transport is
    behind this interface so an offline deployment swaps in LocalRepo
    (= ModelDownloader) unchanged."""

    def __init__(self, base_url: str, cache: ModelDownloader,
                 timeout: float = 60.0):
        self.base_url = base_url.rstrip("/")
        self.cache = cache
        self.timeout = timeout

    def _get(self, path: str) -> bytes:
        import requests
        r = requests.get(f"{self.base_url}/{path}", timeout=self.timeout)
        r.raise_for_status()
        return r.content

    def list_models(self) -> List[ModelSchema]:
        return [ModelSchema(**d) for d in json.loads(self._get("models.json"))]

    def download_by_name(self, name: str) -> ModelSchema:
        for m in self.list_models():
            if m.name != name:
                continue
            local = os.path.join(self.cache.local_path, f"{name}.pt")
            if not (os.path.exists(local)
                    and ModelDownloader._sha256(local) == m.hash):
                blob = self._get(os.path.basename(m.uri))
                if hashlib.sha256(blob).hexdigest() != m.hash:
                    raise IOError(f"hash mismatch downloading {name}")
                with open(local, "wb") as f:
                    f.write(blob)
            m.uri = local
            # register in the local cache index so later offline runs
            # resolve it through ModelDownloader directly
            models = [x for x in self.cache.list_models() if x.name != name]
            models.append(m)
            with open(self.cache._index_path(), "w") as f:
                json.dump([asdict(x) for x in models], f, indent=1)
            return m
        raise KeyError(f"model {name!r} not in remote repo {self.base_url}")

    def load_state(self, name: str):
        return torch.load(self.download_by_name(name).uri, map_location="cpu",
                          weights_only=True)
