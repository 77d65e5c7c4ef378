"""Minibatching stages (core/.../stages/MiniBatchTransformer.scala,
Batchers.scala) + PartitionConsolidator (PartitionConsolidator.scala:23).

A "batch" row holds list-valued cells (the reference's transpose row-batching,
MiniBatchBase:16-41); FlattenBatch undoes it.  DynamicMiniBatch sizes batches
by arrival; TimeIntervalMiniBatch groups by a time window column-less stream
analog — here over an iterator of frames."""
from __future__ import annotations

import queue
import threading
from typing import List

import numpy as np
import pandas as pd

from ..core.param import Param, toInt
from ..core.pipeline import Transformer
from ..core.registry import register


def _batch_frame(df: pd.DataFrame, sizes: List[int]) -> pd.DataFrame:
    rows = []
    s = 0
    for sz in sizes:
        chunk = df.iloc[s:s + sz]
        rows.append({c: list(chunk[c]) for c in df.columns})
        s += sz
    return pd.DataFrame(rows)


@register
class FixedMiniBatchTransformer(Transformer):
    batchSize = Param("batchSize", "rows per batch", 10, toInt)
    maxBufferSize = Param("maxBufferSize", "buffer cap", 2147483647, toInt)

    def _transform(self, df):
        bs = self.get("batchSize")
        n = len(df)
        sizes = [min(bs, n - s) for s in range(0, n, bs)]
        return _batch_frame(df, sizes)


@register
class DynamicMiniBatchTransformer(Transformer):
    """Batch size adapts to arrival rate; in batch mode ≡ one batch
    (DynamicBufferedBatcher, Batchers.scala:12)."""
    maxBatchSize = Param("maxBatchSize", "max rows per batch", 2147483647, toInt)

    def _transform(self, df):
        bs = min(self.get("maxBatchSize"), max(len(df), 1))
        n = len(df)
        sizes = [min(bs, n - s) for s in range(0, n, bs)] or [0]
        if n == 0:
            return df
        return _batch_frame(df, sizes)


@register
class TimeIntervalMiniBatchTransformer(Transformer):
    millisToWait = Param("millisToWait", "batch window ms", 1000, toInt)
    maxBatchSize = Param("maxBatchSize", "max rows per batch", 2147483647, toInt)

    def _transform(self, df):
        # batch-mode semantics: window collects everything available
        bs = min(self.get("maxBatchSize"), max(len(df), 1))
        if len(df) == 0:
            return df
        n = len(df)
        sizes = [min(bs, n - s) for s in range(0, n, bs)]
        return _batch_frame(df, sizes)


@register
class FlattenBatch(Transformer):
    """Invert minibatching: explode all list-valued columns row-aligned."""

    def _transform(self, df):
        rows = []
        for _, row in df.iterrows():
            lens = [len(v) for v in row if isinstance(v, (list, np.ndarray))]
            n = max(lens) if lens else 1
            for i in range(n):
                rows.append({c: (row[c][i] if isinstance(row[c], (list, np.ndarray))
                                 and len(row[c]) > i else row[c])
                             for c in df.columns})
        return pd.DataFrame(rows, columns=df.columns)


@register
class PartitionConsolidator(Transformer):
    """Funnel many partitions' rows through one worker — used to respect
    rate-limited services (PartitionConsolidator.scala:23, Consolidator:52).
    Here partitions are thread shards; rows funnel through one queue-draining
    worker thread to preserve the serialized-consumer semantics."""
    concurrency = Param("concurrency", "consumer threads", 1, toInt)
    timeout = Param("timeout", "seconds to wait", 60, toInt)

    def _transform(self, df):
        q: "queue.Queue" = queue.Queue()
        out_rows = []
        lock = threading.Lock()

        def consumer():
            while True:
                item = q.get()
                if item is None:
                    break
                with lock:
                    out_rows.append(item)

        workers = [threading.Thread(target=consumer)
                   for _ in range(self.get("concurrency"))]
        for w in workers:
            w.start()
        for _, row in df.iterrows():
            q.put(row)
        for _ in workers:
            q.put(None)
        for w in workers:
            w.join(timeout=self.get("timeout"))
        return pd.DataFrame(out_rows, columns=df.columns).reset_index(drop=True)
