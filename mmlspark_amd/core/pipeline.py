"""Estimator / Transformer / Model / Pipeline — the SparkML-shaped API.

``Estimator.fit(df) -> Model``; ``Transformer.transform(df) -> df`` over
pandas DataFrames. Mirrors the reference's SparkML surface (every stage in
SURVEY §1 L2/L4 exposes exactly this) while staying Python-first: a
"DataFrame" is a pandas DataFrame whose feature columns may be numeric
scalars or object columns of fixed-length ``np.ndarray`` vectors (the analog
of SparkML VectorUDT columns).
"""
from __future__ import annotations

import time
from typing import List, Optional

from .param import Param, Params, toList
from .registry import register
from .serialize import load_stage, save_stage
from .telemetry import log_stage_event


class PipelineStage(Params):
    """Common base: uid + params + save/load."""

    def __init__(self, **kwargs):
        super().__init__(**kwargs)
        log_stage_event(self, "constructor")

    # persistence -----------------------------------------------------------
    def save(self, path: str, overwrite: bool = True):
        save_stage(self, path, overwrite=overwrite)
        return self

    write = save

    @classmethod
    def load(cls, path: str):
        stage = load_stage(path)
        if not isinstance(stage, cls):
            raise TypeError(f"loaded {type(stage).__name__}, expected {cls.__name__}")
        return stage

    read = load


class Transformer(PipelineStage):
    def transform(self, df):
        # pyarrow Tables / pyspark DataFrames accepted everywhere; the
        # output comes back in the caller's kind (core/interop.py)
        from .interop import coerce_input, restore_output
        df, kind = coerce_input(df)
        t0 = time.time()
        try:
            out = self._transform(df)
        except Exception as e:  # telemetry parity: error path logging
            log_stage_event(self, "transform", error=repr(e))
            raise
        log_stage_event(self, "transform", ms=(time.time() - t0) * 1e3)
        return restore_output(out, kind)

    def _transform(self, df):
        raise NotImplementedError

    def __call__(self, df):
        return self.transform(df)


class Estimator(PipelineStage):
    def fit(self, df, params: Optional[dict] = None):
        from .interop import coerce_input
        df, _ = coerce_input(df)
        inst = self.copy(params) if params else self
        t0 = time.time()
        try:
            model = inst._fit(df)
        except Exception as e:
            log_stage_event(self, "fit", error=repr(e))
            raise
        log_stage_event(self, "fit", ms=(time.time() - t0) * 1e3)
        return model

    def _fit(self, df) -> "Model":
        raise NotImplementedError


class Model(Transformer):
    """A fitted Transformer produced by an Estimator."""


@register
class Pipeline(Estimator):
    """Sequential stages; fits estimators in order, collects a PipelineModel.

    Analog of org.apache.spark.ml.Pipeline as used throughout the reference
    (e.g. SimpleHTTPTransformer.makePipeline, core/.../io/http/SimpleHTTPTransformer.scala:114).
    """

    stages = Param("stages", "pipeline stages", default=None, converter=toList,
                   is_complex=True)

    def __init__(self, stages: Optional[List[PipelineStage]] = None, **kwargs):
        super().__init__(**kwargs)
        if stages is not None:
            self.set("stages", stages)

    def _fit(self, df):
        fitted: List[Transformer] = []
        cur = df
        stages = self.get("stages") or []
        for st in stages:
            if isinstance(st, Estimator):
                m = st.fit(cur)
                fitted.append(m)
                cur = m.transform(cur)
            elif isinstance(st, Transformer):
                fitted.append(st)
                cur = st.transform(cur)
            else:
                raise TypeError(f"stage {st!r} is neither Estimator nor Transformer")
        return PipelineModel(stages=fitted)


@register
class PipelineModel(Model):
    stages = Param("stages", "fitted stages", default=None, converter=toList,
                   is_complex=True)

    def __init__(self, stages: Optional[List[Transformer]] = None, **kwargs):
        super().__init__(**kwargs)
        if stages is not None:
            self.set("stages", stages)

    def _transform(self, df):
        cur = df
        for st in self.get("stages") or []:
            cur = st.transform(cur)
        return cur
