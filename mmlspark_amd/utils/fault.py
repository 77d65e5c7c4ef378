"""Fault-tolerance + timing utilities.

Analogs of the reference's core/utils toolbox: FaultToleranceUtils.
retryWithTimeout (core/.../core/utils/FaultToleranceUtils.scala:33 — wraps
native init and network rendezvous), StopWatch (core/utils/StopWatch.scala),
AsyncUtils bounded-concurrency map, StreamUtilities.using, and the
SharedVariable per-process singleton pool (io/http/SharedVariable.scala:18).
"""
from __future__ import annotations

import threading
import time
from concurrent.futures import ThreadPoolExecutor, TimeoutError as _FTimeout
from contextlib import contextmanager
from typing import Any, Callable, Dict, Iterable, List, Optional, TypeVar

T = TypeVar("T")


def retry_with_timeout(fn: Callable[[], T], timeout_s: float,
                       retries: int = 3, backoff_s: float = 0.1) -> T:
    """Run fn with a wall-clock timeout, retrying with exponential backoff
    (FaultToleranceUtils.retryWithTimeout semantics: a hang counts as a
    failure, the last error propagates)."""
    err: Optional[BaseException] = None
    for attempt in range(max(1, retries)):
        with ThreadPoolExecutor(max_workers=1) as ex:
            fut = ex.submit(fn)
            try:
                return fut.result(timeout=timeout_s)
            except _FTimeout as e:
                fut.cancel()
                err = TimeoutError(f"timed out after {timeout_s}s") \
                    .with_traceback(e.__traceback__)
            except Exception as e:  # noqa: BLE001 — retry any failure
                err = e
        if attempt + 1 < retries:
            time.sleep(backoff_s * (2 ** attempt))
    raise err  # type: ignore[misc]


class StopWatch:
    """Accumulating stopwatch (StopWatch.scala): measure {} blocks."""

    def __init__(self):
        self.elapsed_s = 0.0
        self._t0: Optional[float] = None

    def start(self):
        self._t0 = time.perf_counter()

    def stop(self):
        if self._t0 is not None:
            self.elapsed_s += time.perf_counter() - self._t0
            self._t0 = None

    def restart(self):
        self.elapsed_s = 0.0
        self.start()

    @contextmanager
    def measure(self):
        self.start()
        try:
            yield self
        finally:
            self.stop()


def async_map(fn: Callable[[Any], T], items: Iterable[Any],
              concurrency: int = 8,
              timeout_s: Optional[float] = None) -> List[T]:
    """Bounded-concurrency map preserving order (AsyncUtils analog — the
    buffered-future pipeline of io/http/Clients.scala AsyncClient)."""
    with ThreadPoolExecutor(max_workers=max(1, concurrency)) as ex:
        futs = [ex.submit(fn, it) for it in items]
        return [f.result(timeout=timeout_s) for f in futs]


@contextmanager
def using(*resources):
    """StreamUtilities.using: close every resource on exit, first error wins."""
    try:
        yield resources if len(resources) > 1 else resources[0]
    finally:
        for r in reversed(resources):
            close = getattr(r, "close", None)
            if callable(close):
                try:
                    close()
                except Exception:  # noqa: BLE001 — best-effort cleanup
                    pass


class SharedVariable:
    """Lazily-constructed per-process singleton (SharedVariable.scala:18):
    one instance per construction site shared across threads, e.g. one HTTP
    client pool per process."""

    _registry: Dict[int, Any] = {}
    _lock = threading.Lock()

    def __init__(self, factory: Callable[[], T]):
        self._factory = factory
        self._key = id(self)

    def get(self) -> T:
        reg = SharedVariable._registry
        if self._key not in reg:
            with SharedVariable._lock:
                if self._key not in reg:
                    reg[self._key] = self._factory()
        return reg[self._key]

    def set(self, value: T):
        with SharedVariable._lock:
            SharedVariable._registry[self._key] = value
