"""Reflective per-stage fuzzing over the FULL registry — the analog of the
reference's cross-module meta-suite (core/.../core/test/fuzzing/
FuzzingTest.scala:27,35,82) + SerializationFuzzing (Fuzzing.scala:222):
every registered stage must construct, carry documented params, and
save→load→param-equality round-trip through the metadata.json format.
"""
import numpy as np
import pytest

import mmlspark_amd
from mmlspark_amd.core.registry import all_stages
from mmlspark_amd.core.serialize import load_stage, save_stage

mmlspark_amd._register_all()
STAGES = sorted(all_stages().items())

# stages whose __init__ requires real fitted state (model side of an
# estimator, wrappers around live objects) — the reference's equivalents get
# hand-built testObjects; here they are exercised by their own module tests
REQUIRES_STATE = set()


def _construct(cls):
    try:
        return cls()
    except Exception:
        return None


def test_registry_is_large_and_unique():
    names = [n for n, _ in STAGES]
    assert len(names) == len(set(names))
    assert len(names) >= 140  # full surface registered (FuzzingTest.scala:27)


@pytest.mark.parametrize("name,cls", STAGES, ids=[n for n, _ in STAGES])
def test_stage_constructs_and_has_docs(name, cls):
    assert (cls.__doc__ or "").strip() or True  # class doc optional
    for p in cls.params().values():
        assert p.doc and p.doc.strip(), f"{name}.{p.name} has no doc"
        assert p.name[0].islower(), f"{name}.{p.name} not camelCase"
    obj = _construct(cls)
    if obj is None:
        assert name in REQUIRES_STATE, \
            f"{name} not constructible with defaults and not whitelisted"


@pytest.mark.parametrize("name,cls", STAGES, ids=[n for n, _ in STAGES])
def test_stage_serialization_roundtrip(name, cls, tmp_path):
    obj = _construct(cls)
    if obj is None:
        pytest.skip("requires fitted state")
    # perturb one simple param so the round-trip moves real data
    for p in obj.params().values():
        if isinstance(p.default, bool):
            obj.set(p.name, not p.default)
            break
        if isinstance(p.default, (int, float)) and not isinstance(p.default, bool):
            obj.set(p.name, p.default + 1)
            break
        if isinstance(p.default, str) and p.default:
            obj.set(p.name, p.default + "_x")
            break
    path = str(tmp_path / name)
    save_stage(obj, path)
    back = load_stage(path)
    assert type(back) is type(obj)
    for p in obj.params().values():
        if p.is_complex:
            continue
        a, b = obj.get(p.name), back.get(p.name)
        if isinstance(a, np.ndarray):
            np.testing.assert_array_equal(a, b)
        else:
            assert a == b, f"{name}.{p.name}: {a!r} != {b!r}"


def test_setter_getter_generation():
    """Generated setX/getX accessors exist for every param (Wrappable
    analog, Wrappable.scala:180-230)."""
    checked = 0
    for name, cls in STAGES:
        obj = _construct(cls)
        if obj is None:
            continue
        for p in obj.params().values():
            camel = p.name[0].upper() + p.name[1:]
            assert callable(getattr(obj, f"set{camel}")), (name, p.name)
            assert callable(getattr(obj, f"get{camel}")), (name, p.name)
            checked += 1
    assert checked > 500
