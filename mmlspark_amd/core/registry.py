"""Stage registry — single source of truth for every public stage class.

Serves the roles of the reference's jar-reflection (``JarLoadingUtils
.instantiateServices``, core/.../core/utils/JarLoadingUtils.scala) used by both
codegen (§2.7 of SURVEY) and the cross-module fuzzing meta-suite
(core/.../core/test/fuzzing/FuzzingTest.scala): tests iterate every registered
stage and assert serialization round-trips.
"""
from __future__ import annotations

from typing import Dict, Type

_REGISTRY: Dict[str, Type] = {}


def register(cls):
    """Class decorator: make a stage discoverable by name for load() + fuzzing."""
    _REGISTRY[cls.__name__] = cls
    _REGISTRY[f"{cls.__module__}.{cls.__name__}"] = cls
    return cls


def lookup(name: str):
    if name in _REGISTRY:
        return _REGISTRY[name]
    # fall back: qualified name whose tail matches
    tail = name.rsplit(".", 1)[-1]
    if tail in _REGISTRY:
        return _REGISTRY[tail]
    raise KeyError(f"stage class {name!r} is not registered")


def all_stages() -> Dict[str, Type]:
    """Unique registered classes keyed by bare class name."""
    return {k: v for k, v in _REGISTRY.items() if "." not in k}
