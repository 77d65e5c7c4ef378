"""Typed Param system — the single config system of the framework.

Mirrors the role of SparkML ``Params`` in the reference (every knob on every
stage is a typed, documented param serialized into model metadata; cf.
reference ``core/src/main/scala/org/apache/spark/ml/param/*`` and
``lightgbm/src/main/scala/.../params/LightGBMParams.scala``).  Python-first:
params are declared as class attributes; setters/getters are generated.
"""
from __future__ import annotations

import copy
import uuid
from typing import Any, Callable, Dict, Optional


class Param:
    """One typed parameter: name, doc, default, optional converter/validator."""

    __slots__ = ("name", "doc", "default", "converter", "is_complex")

    def __init__(self, name: str, doc: str = "", default: Any = None,
                 converter: Optional[Callable[[Any], Any]] = None,
                 is_complex: bool = False):
        self.name = name
        self.doc = doc
        self.default = default
        self.converter = converter
        # complex params are persisted as side files, not JSON metadata
        # (analog of reference ComplexParam, core/.../core/serialize/ComplexParam.scala:13)
        self.is_complex = is_complex

    def convert(self, value: Any) -> Any:
        return self.converter(value) if self.converter is not None else value

    def __repr__(self):
        return f"Param({self.name!r})"


def _identity(x):
    return x


# common converters
def toInt(x):
    return int(x)


def toFloat(x):
    return float(x)


def toBool(x):
    return bool(x)


def toString(x):
    return str(x)


def toList(x):
    return list(x)


class Params:
    """Base for anything carrying Params. Declares params as class attributes.

    Subclasses declare ``myParam = Param("myParam", "doc", default)``.
    Instances get ``set(param | name, value)``, ``get``, ``setParams(**kw)``,
    generated ``setMyParam/getMyParam`` via __getattr__ fallback.
    """

    def __init__(self, **kwargs):
        self.uid = f"{type(self).__name__}_{uuid.uuid4().hex[:12]}"
        self._paramMap: Dict[str, Any] = {}
        if kwargs:
            self.setParams(**kwargs)

    # ---- declaration discovery -------------------------------------------------
    @classmethod
    def params(cls) -> Dict[str, Param]:
        out: Dict[str, Param] = {}
        for klass in reversed(cls.__mro__):
            for k, v in vars(klass).items():
                if isinstance(v, Param):
                    out[v.name] = v
        return out

    def param(self, name: str) -> Param:
        p = self.params().get(name)
        if p is None:
            raise KeyError(f"{type(self).__name__} has no param {name!r}")
        return p

    def hasParam(self, name: str) -> bool:
        return name in self.params()

    # ---- set/get ---------------------------------------------------------------
    def set(self, param, value):
        p = param if isinstance(param, Param) else self.param(param)
        self._paramMap[p.name] = p.convert(value)
        return self

    def get(self, param):
        p = param if isinstance(param, Param) else self.param(param)
        if p.name in self._paramMap:
            return self._paramMap[p.name]
        return p.default

    def isSet(self, param) -> bool:
        p = param if isinstance(param, Param) else self.param(param)
        return p.name in self._paramMap

    def setParams(self, **kwargs):
        for k, v in kwargs.items():
            if v is None and not self.hasParam(k):
                continue
            self.set(k, v)
        return self

    def explainParams(self) -> str:
        lines = []
        for name, p in sorted(self.params().items()):
            cur = self._paramMap.get(name, p.default)
            lines.append(f"{name}: {p.doc} (default: {p.default!r}, current: {cur!r})")
        return "\n".join(lines)

    def extractParamMap(self) -> Dict[str, Any]:
        out = {name: p.default for name, p in self.params().items()}
        out.update(self._paramMap)
        return out

    def copy(self, extra: Optional[Dict[str, Any]] = None):
        c = copy.deepcopy(self)
        c.uid = self.uid
        if extra:
            c.setParams(**extra)
        return c

    # ---- generated accessors ----------------------------------------------------
    def __getattr__(self, item: str):
        # only called when normal lookup fails
        if item.startswith("set") and len(item) > 3:
            name = item[3].lower() + item[4:]
            if self.hasParam(name):
                def setter(value, _name=name):
                    return self.set(_name, value)
                return setter
        if item.startswith("get") and len(item) > 3:
            name = item[3].lower() + item[4:]
            if self.hasParam(name):
                return lambda _name=name: self.get(_name)
        raise AttributeError(f"{type(self).__name__} has no attribute {item!r}")

    def __repr__(self):
        return f"{type(self).__name__}(uid={self.uid})"
