"""Elastic gang restart — the training-job fault-tolerance analog of
Spark's barrier task retry (the reference's elastic story is
"Spark-native task retry + barrier gang restart").

run_elastic relaunches a whole training command (typically a torchrun
gang) until it exits cleanly; combined with train_booster's
iteration-level checkpoints the restarted gang resumes mid-training
instead of from scratch."""
from __future__ import annotations

import subprocess
import time
from typing import List, Optional


def run_elastic(argv: List[str], max_restarts: int = 2,
                env: Optional[dict] = None, backoff_s: float = 1.0,
                timeout: Optional[float] = None) -> int:
    """Run `argv`; on failure restart (gang restart). Returns the number of
    restarts that were needed. Raises after max_restarts failures."""
    last_rc = None
    for attempt in range(max_restarts + 1):
        rc = subprocess.call(argv, env=env, timeout=timeout)
        if rc == 0:
            return attempt
        last_rc = rc
        time.sleep(backoff_s * (attempt + 1))
    raise RuntimeError(
        f"command failed after {max_restarts + 1} attempts (rc={last_rc})")
