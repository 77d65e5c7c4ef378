"""HTTP-on-DataFrame client stack (core/.../io/http parity).

HTTPTransformer (HTTPTransformer.scala:88): request-struct column → response
column, with a shared client per process (SharedVariable analog), concurrent
async calls, and the advanced retry/backoff handler incl. 429 Retry-After
(HTTPClients.HandlingUtils.advanced:75-121).
SimpleHTTPTransformer (SimpleHTTPTransformer.scala:65): minibatch →
JSONInputParser → HTTPTransformer → error split → JSONOutputParser → flatten,
composed as an internal PipelineModel (makePipeline:114-157)."""
from __future__ import annotations

import concurrent.futures as cf
import json
import time

import numpy as np
import pandas as pd

from ..core.param import Param, toBool, toInt, toString
from ..core.pipeline import PipelineModel, Transformer
from ..core.registry import register
from .http_schema import HTTPRequestData, HTTPResponseData

_SESSION = None


def _session():
    global _SESSION
    if _SESSION is None:
        import requests
        _SESSION = requests.Session()
    return _SESSION


def advanced_handler(req: HTTPRequestData, retries=(100, 500, 1000),
                     timeout: float = 60.0) -> HTTPResponseData:
    """Send with retry/backoff; honors 429 Retry-After (HTTPClients.scala:75-121)."""
    last = None
    for attempt in range(len(retries) + 1):
        try:
            r = _session().request(
                req.method, req.url, headers=req.headers, data=req.entity,
                timeout=timeout)
            if r.status_code == 429 and attempt < len(retries):
                wait = float(r.headers.get("Retry-After",
                                           retries[attempt] / 1000.0))
                time.sleep(min(wait, 30.0))
                continue
            if 500 <= r.status_code < 600 and attempt < len(retries):
                time.sleep(retries[attempt] / 1000.0)
                continue
            return HTTPResponseData(r.status_code, r.reason,
                                    dict(r.headers), r.content)
        except Exception as e:  # connection errors retry then surface as 502
            last = e
            if attempt < len(retries):
                time.sleep(retries[attempt] / 1000.0)
    return HTTPResponseData(502, f"client error: {last!r}", {}, None)


@register
class HTTPTransformer(Transformer):
    inputCol = Param("inputCol", "HTTPRequestData column", "request")
    outputCol = Param("outputCol", "HTTPResponseData column", "response")
    concurrency = Param("concurrency", "parallel in-flight requests", 1, toInt)
    timeout = Param("timeout", "per-request timeout seconds", 60.0)

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        reqs = df[self.get("inputCol")].tolist()
        conc = self.get("concurrency")
        tmo = float(self.get("timeout"))
        if conc > 1:
            with cf.ThreadPoolExecutor(conc) as pool:
                resps = list(pool.map(
                    lambda r: advanced_handler(r, timeout=tmo) if r is not None
                    else None, reqs))
        else:
            resps = [advanced_handler(r, timeout=tmo) if r is not None else None
                     for r in reqs]
        out = df.copy()
        out[self.get("outputCol")] = resps
        return out


@register
class JSONInputParser(Transformer):
    """Row (minus excluded cols) → JSON POST HTTPRequestData (Parsers.scala:35)."""
    inputCol = Param("inputCol", "column holding the JSON payload dict", None)
    outputCol = Param("outputCol", "request column", "request")
    url = Param("url", "target url", "")
    method = Param("method", "http method", "POST", toString)
    headers = Param("headers", "extra headers", None)

    def _transform(self, df):
        headers = {"Content-Type": "application/json"}
        headers.update(self.get("headers") or {})
        out = df.copy()
        col = self.get("inputCol")

        def mk(payload):
            body = json.dumps(payload, default=_np_default).encode()
            return HTTPRequestData(url=self.get("url"),
                                   method=self.get("method"),
                                   headers=dict(headers), entity=body)
        if col:
            out[self.get("outputCol")] = [mk(v) for v in df[col]]
        else:
            out[self.get("outputCol")] = [mk(r._asdict() if hasattr(r, "_asdict")
                                             else dict(r))
                                          for r in df.to_dict("records")]
        return out


def _np_default(o):
    if isinstance(o, np.ndarray):
        return o.tolist()
    if isinstance(o, (np.floating, np.integer)):
        return o.item()
    raise TypeError(type(o))


@register
class JSONOutputParser(Transformer):
    inputCol = Param("inputCol", "response column", "response")
    outputCol = Param("outputCol", "parsed column", "parsed")

    def _transform(self, df):
        out = df.copy()
        out[self.get("outputCol")] = [
            r.json() if r is not None and r.entity else None
            for r in df[self.get("inputCol")]]
        return out


@register
class StringOutputParser(Transformer):
    inputCol = Param("inputCol", "response column", "response")
    outputCol = Param("outputCol", "string column", "parsed")

    def _transform(self, df):
        out = df.copy()
        out[self.get("outputCol")] = [r.text if r is not None else None
                                      for r in df[self.get("inputCol")]]
        return out


@register
class CustomInputParser(Transformer):
    inputCol = Param("inputCol", "input column", None)
    outputCol = Param("outputCol", "request column", "request")

    def __init__(self, udf=None, **kwargs):
        super().__init__(**kwargs)
        self._udf = udf

    def setUDF(self, fn):
        self._udf = fn
        return self

    def _transform(self, df):
        out = df.copy()
        out[self.get("outputCol")] = [self._udf(v)
                                      for v in df[self.get("inputCol")]]
        return out


@register
class CustomOutputParser(Transformer):
    inputCol = Param("inputCol", "response column", "response")
    outputCol = Param("outputCol", "parsed column", "parsed")

    def __init__(self, udf=None, **kwargs):
        super().__init__(**kwargs)
        self._udf = udf

    def setUDF(self, fn):
        self._udf = fn
        return self

    def _transform(self, df):
        out = df.copy()
        out[self.get("outputCol")] = [self._udf(v)
                                      for v in df[self.get("inputCol")]]
        return out


@register
class DropHTTPErrors(Transformer):
    """Error-split stage: keep 2xx rows, collect the rest (SimpleHTTPTransformer
    error column semantics)."""
    inputCol = Param("inputCol", "response column", "response")
    errorCol = Param("errorCol", "error output column", "errors")

    def _transform(self, df):
        out = df.copy()
        out[self.get("errorCol")] = [
            None if (r is not None and 200 <= r.statusCode < 300)
            else (r.to_dict() if r is not None else {"statusCode": -1})
            for r in df[self.get("inputCol")]]
        return out


@register
class SimpleHTTPTransformer(Transformer):
    inputCol = Param("inputCol", "payload column", None)
    outputCol = Param("outputCol", "parsed output column", "output")
    url = Param("url", "target url", "")
    concurrency = Param("concurrency", "parallel requests", 1, toInt)
    flattenOutputBatches = Param("flattenOutputBatches", "flatten", True, toBool)
    errorCol = Param("errorCol", "error column", "errors")

    def _make_pipeline(self) -> PipelineModel:
        return PipelineModel(stages=[
            JSONInputParser(inputCol=self.get("inputCol"), outputCol="__req",
                            url=self.get("url")),
            HTTPTransformer(inputCol="__req", outputCol="__resp",
                            concurrency=self.get("concurrency")),
            DropHTTPErrors(inputCol="__resp", errorCol=self.get("errorCol")),
            JSONOutputParser(inputCol="__resp",
                             outputCol=self.get("outputCol")),
            __import__("mmlspark_amd.stages.basic", fromlist=["DropColumns"])
            .DropColumns(cols=["__req", "__resp"]),
        ])

    def _transform(self, df):
        return self._make_pipeline().transform(df)
