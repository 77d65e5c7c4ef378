"""Functional tests for stages previously covered only by the reflective
fuzzing suite: translator payloads, custom parsers, batching variants,
RankingAdapter, SuperpixelTransformer, IndexToValue."""
import numpy as np
import pandas as pd


def test_translator_family_payloads():
    from mmlspark_amd.io_http.cognitive import (BreakSentence, Detect,
                                                Translate, Transliterate)
    row = pd.Series({"text": "hello world"})
    t = Translate(url="http://x/", toLanguage=["de", "fr"])
    assert t._payload(row) == [{"Text": "hello world"}]
    assert t._row_url(row) == "http://x/?to=de&to=fr"
    assert Detect(url="http://x/")._row_url(row) == "http://x/"
    assert BreakSentence(url="http://x/")._row_url(row) == "http://x/"
    tr = Transliterate(url="http://x/", fromScript="Latn", toScript="Cyrl")
    assert tr.get("toScript") == "Cyrl"


def test_image_service_urls():
    from mmlspark_amd.io_http.cognitive import AnalyzeImage, OCR
    row = pd.Series({"url_col": "http://img/x.png"})
    a = AnalyzeImage(url="http://svc/analyze", imageUrlCol="url_col",
                     visualFeatures=["Categories", "Tags"])
    assert a._payload(row) == {"url": "http://img/x.png"}
    assert "visualFeatures=Categories,Tags" in a._row_url(row)
    assert OCR(url="http://svc/ocr", imageUrlCol="url_col")._payload(row) \
        == {"url": "http://img/x.png"}


def test_custom_and_string_parsers():
    from mmlspark_amd.io_http.client import (CustomInputParser,
                                             CustomOutputParser,
                                             StringOutputParser)
    from mmlspark_amd.io_http.http_schema import HTTPResponseData

    df = pd.DataFrame({"q": ["a", "b"]})
    cip = CustomInputParser(inputCol="q", udf=lambda v: {"query": v.upper()})
    reqs = cip.transform(df)["request"].tolist()
    assert reqs[0] == {"query": "A"}

    resp = HTTPResponseData(statusCode=200, entity=b'{"ok":1}')
    rdf = pd.DataFrame({"response": [resp, None]})
    sop = StringOutputParser(inputCol="response", outputCol="s")
    ss = sop.transform(rdf)["s"].tolist()
    assert ss[0] == '{"ok":1}' and ss[1] is None

    cop = CustomOutputParser(inputCol="response", outputCol="code",
                             udf=lambda r: None if r is None
                             else r.statusCode)
    codes = cop.transform(rdf)["code"].tolist()
    assert codes[0] == 200 and (codes[1] is None or np.isnan(codes[1]))


def test_time_interval_batcher_and_repartition():
    from mmlspark_amd.stages.basic import Repartition
    from mmlspark_amd.stages.batching import TimeIntervalMiniBatchTransformer
    df = pd.DataFrame({"x": list(range(10))})
    out = TimeIntervalMiniBatchTransformer(maxBatchSize=4).transform(df)
    lens = [len(b) for b in out["x"]]
    assert lens == [4, 4, 2]
    assert Repartition(n=8).transform(df).equals(df)


def test_index_to_value_roundtrip():
    from mmlspark_amd.stages.featurize import IndexToValue, ValueIndexer
    df = pd.DataFrame({"c": ["red", "blue", "red", "green"]})
    m = ValueIndexer(inputCol="c", outputCol="i").fit(df)
    idx = m.transform(df)
    back = IndexToValue(inputCol="i", outputCol="c2",
                        levels=m.get("levels")).transform(idx)
    assert back["c2"].tolist() == df["c"].tolist()


def test_ranking_adapter_feeds_evaluator():
    from mmlspark_amd.models.sar import SAR, RankingAdapter, RankingEvaluator
    rng = np.random.default_rng(3)
    n_users, n_items = 60, 30
    rows = []
    for u in range(n_users):
        liked = rng.choice(n_items // 2, size=6, replace=False) \
            if u % 2 == 0 else rng.choice(
                np.arange(n_items // 2, n_items), size=6, replace=False)
        for it in liked:
            rows.append((u, int(it), 5.0))
    df = pd.DataFrame(rows, columns=["userIdx", "itemIdx", "rating"])
    model = SAR(userCol="userIdx", itemCol="itemIdx",
                ratingCol="rating").fit(df)
    adapted = RankingAdapter(recommenderModel=model, k=10).transform(df)
    assert {"prediction", "label"} <= set(adapted.columns)
    ev = RankingEvaluator(k=10)
    ndcg = ev.evaluate(adapted)
    assert 0.2 < ndcg <= 1.0  # strong block structure → well above random


def test_superpixel_transformer_stage():
    from mmlspark_amd.explainers.lime import SuperpixelTransformer
    img = np.zeros((32, 32, 3), np.uint8)
    img[:, 16:] = 255
    out = SuperpixelTransformer(cellSize=8).transform(
        pd.DataFrame({"image": [img]}))
    segs = out["superpixels"].iloc[0]
    assert segs.shape == (32, 32)
    assert len(np.unique(segs)) >= 4  # multiple segments found


def test_backend_fails_loudly_without_extension(monkeypatch):
    """On a GPU tensor a missing HIP extension must raise, never silently
    fall back to eager torch (round-end native-load check contract)."""
    import mmlspark_amd.ops.backend as B
    monkeypatch.setattr(B, "_EXT", False)
    monkeypatch.setattr(B, "_EXT_ERR", "simulated")
    import pytest as _pytest
    with _pytest.raises(RuntimeError, match="HIP extension is required"):
        B._require_ext()


def test_default_device_selection():
    import torch
    from mmlspark_amd.utils.devices import default_device
    assert default_device("cpu").type == "cpu"
    auto = default_device("auto")
    assert auto.type == ("cuda" if torch.cuda.is_available() else "cpu")
    assert default_device(None).type == auto.type
