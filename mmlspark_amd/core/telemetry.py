"""Usage telemetry — JSON event per constructor/fit/transform (+error path).

Analog of the reference's BasicLogging (core/.../logging/BasicLogging.scala:26-71):
every stage logs {uid, className, method, buildVersion} on lifecycle events.
Here: emitted to the ``mmlspark_amd.telemetry`` logger at DEBUG (off by
default), and appended to an in-memory ring for tests/observability.
"""
from __future__ import annotations

import collections
import json
import logging

from .. import __version__ as _BUILD_VERSION

logger = logging.getLogger("mmlspark_amd.telemetry")

_EVENTS = collections.deque(maxlen=1024)


def log_stage_event(stage, method: str, **extra):
    evt = {
        "uid": getattr(stage, "uid", "?"),
        "className": type(stage).__name__,
        "method": method,
        "buildVersion": _BUILD_VERSION,
    }
    evt.update(extra)
    _EVENTS.append(evt)
    if logger.isEnabledFor(logging.DEBUG):
        logger.debug(json.dumps(evt))


def recent_events():
    return list(_EVENTS)
