"""Process-wide shared state (io/http/SharedVariable.scala parity).

The reference shares non-serializable objects (HTTP clients, native handles)
across tasks inside one executor JVM via a UUID-keyed TrieMap pool
(SharedVariable.scala:18, SharedSingleton:36).  The MI355X runtime is one
process per GPU, so the pool is a plain module-level dict guarded by a lock;
values are built lazily by the constructor thunk and survive pickling as the
key alone (rebuilt on first access in the new process — the same semantics
the reference gets from transient + lazy re-init after task deserialization).
"""
from __future__ import annotations

import threading
import uuid
from typing import Callable, Generic, TypeVar

T = TypeVar("T")

_POOL: dict = {}
_LOCK = threading.Lock()


class SharedVariable(Generic[T]):
    """Lazily-constructed per-process shared value.

    >>> client = SharedVariable(lambda: make_session())
    >>> client.get() is client.get()   # same object, built once
    True
    """

    def __init__(self, ctor: Callable[[], T], key: str | None = None):
        self._ctor = ctor
        self._key = key or uuid.uuid4().hex

    def get(self) -> T:
        with _LOCK:
            if self._key not in _POOL:
                _POOL[self._key] = self._ctor()
            return _POOL[self._key]

    def set(self, value: T) -> None:
        with _LOCK:
            _POOL[self._key] = value

    def __getstate__(self):  # the value itself never travels
        return {"_key": self._key, "_ctor": self._ctor}

    def __setstate__(self, state):
        self.__dict__.update(state)


class SharedSingleton(SharedVariable[T]):
    """SharedVariable keyed by the constructor's identity: every instance
    built from the same ctor in this process shares ONE value
    (SharedVariable.scala SharedSingleton:36)."""

    def __init__(self, ctor: Callable[[], T]):
        key = f"singleton:{getattr(ctor, '__module__', '?')}." \
              f"{getattr(ctor, '__qualname__', repr(ctor))}"
        super().__init__(ctor, key=key)


def clear_pool() -> None:
    """Test hook: drop every shared value in this process."""
    with _LOCK:
        _POOL.clear()
