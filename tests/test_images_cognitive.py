"""ImageTransformer / ImageSetAugmenter + cognitive HTTP transformers
(against local mock services)."""
import json

import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.models.images import ImageSetAugmenter, ImageTransformer
from mmlspark_amd.serving.server import ServingServer


def _img(h=20, w=30):
    rng = np.random.default_rng(0)
    return rng.integers(0, 255, size=(h, w, 3)).astype(np.uint8)


def test_image_transformer_pipeline():
    df = pd.DataFrame({"image": [_img()]})
    t = (ImageTransformer()
         .resize(10, 12)
         .crop(2, 2, 6, 8)
         .flip(1)
         .blur(3, 3))
    out = t.transform(df)
    res = out["out_image"].iloc[0]
    assert res.shape == (6, 8, 3)


def test_image_transformer_ops():
    img = _img()
    df = pd.DataFrame({"image": [img]})
    gray = ImageTransformer().colorFormat("gray").transform(df)["out_image"].iloc[0]
    assert gray.shape == (20, 30, 1)
    th = ImageTransformer().threshold(128).transform(df)["out_image"].iloc[0]
    assert set(np.unique(th)).issubset({0, 255})
    g = ImageTransformer().gaussianKernel(5, 1.5).transform(df)["out_image"].iloc[0]
    assert g.shape == img.shape
    flipped = ImageTransformer().flip(1).transform(df)["out_image"].iloc[0]
    np.testing.assert_array_equal(flipped, img[:, ::-1])


def test_image_set_augmenter():
    df = pd.DataFrame({"image": [_img(), _img()]})
    out = ImageSetAugmenter(flipLeftRight=True, flipUpDown=True).transform(df)
    assert len(out) == 6


def test_image_transformer_save_load(tmp_path):
    import os
    t = ImageTransformer().resize(8, 8).threshold(100)
    t.save(os.path.join(tmp_path, "it"))
    t2 = ImageTransformer.load(os.path.join(tmp_path, "it"))
    df = pd.DataFrame({"image": [_img()]})
    np.testing.assert_array_equal(t.transform(df)["out_image"].iloc[0],
                                  t2.transform(df)["out_image"].iloc[0])


# ----------------------------------------------------------------- cognitive
@pytest.fixture(scope="module")
def mock_service():
    """Local mock that echoes a sentiment-like response."""
    def handler(payloads):
        out = []
        for p in payloads:
            docs = p.get("documents", [])
            out.append({"documents": [
                {"id": d["id"], "sentiment":
                 "positive" if "good" in d.get("text", "") else "negative"}
                for d in docs], "errors": []})
        return out
    srv = ServingServer(handler, port=0, mode="continuous").start()
    yield f"http://127.0.0.1:{srv.port}/"
    srv.stop()


def test_text_sentiment_against_mock(mock_service):
    from mmlspark_amd.io_http.cognitive import TextSentiment
    df = pd.DataFrame({"text": ["good day", "awful day"],
                       "key": ["k1", "k2"]})
    ts = TextSentiment(url=mock_service, subscriptionKeyCol="key",
                       textCol="text", concurrency=2)
    out = ts.transform(df)
    r0 = out["response"].iloc[0]
    assert r0["documents"][0]["sentiment"] == "positive"
    r1 = out["response"].iloc[1]
    assert r1["documents"][0]["sentiment"] == "negative"
    assert out["errors"].isna().all()


def test_cognitive_error_column():
    from mmlspark_amd.io_http.cognitive import TextSentiment
    # unreachable endpoint -> error column populated, no exception
    df = pd.DataFrame({"text": ["x"]})
    ts = TextSentiment(url="http://127.0.0.1:1/", textCol="text", timeout=0.5)
    out = ts.transform(df)
    assert out["errors"].iloc[0] is not None


def test_anomaly_detector_shape(mock_service):
    from mmlspark_amd.io_http.cognitive import DetectLastAnomaly
    series = [{"timestamp": f"2020-01-{i+1:02d}T00:00:00Z", "value": float(i)}
              for i in range(12)]
    df = pd.DataFrame({"series": [series]})
    det = DetectLastAnomaly(url=mock_service, seriesCol="series",
                            granularity="daily")
    out = det.transform(df)
    assert out["response"].iloc[0] is not None


def test_bing_image_search_get(mock_service):
    from mmlspark_amd.io_http.cognitive import BingImageSearch
    df = pd.DataFrame({"q": ["cats"]})
    b = BingImageSearch(url=mock_service, qCol="q", count=3)
    out = b.transform(df)
    # mock returns a JSON body for GET-with-params too (echo handler tolerates)
    assert "response" in out.columns


def test_simple_detect_anomalies_grouped(mock_service):
    """SimpleDetectAnomalies (AnomalyDetection.scala): one request per
    group, response joined back onto every member row."""
    from mmlspark_amd.io_http.cognitive import SimpleDetectAnomalies
    df = pd.DataFrame({
        "group": ["a", "a", "a", "b", "b"],
        "timestamp": [f"2024-01-0{i}T00:00:00Z" for i in (1, 2, 3, 1, 2)],
        "value": [1.0, 2.0, 50.0, 5.0, 5.5],
    })
    det = SimpleDetectAnomalies(url=mock_service, outputCol="anom")
    out = det.transform(df)
    assert len(out) == 5
    # all rows of one group share the same (single) service response
    a = out[out["group"] == "a"]["anom"].tolist()
    assert all(r == a[0] for r in a)
    assert out["anom"].notna().all()


def test_get_custom_model_uses_get(mock_service):
    from mmlspark_amd.io_http.cognitive import GetCustomModel, ListCustomModels
    df = pd.DataFrame({"x": [1]})
    g = GetCustomModel(url=mock_service + "models/{modelId}", modelId="m-7",
                       outputCol="meta")
    out = g.transform(df)
    assert out["meta"].iloc[0] is not None
    ls = ListCustomModels(url=mock_service, outputCol="models")
    assert ls.transform(df)["models"].iloc[0] is not None


def test_dictionary_examples_payload():
    from mmlspark_amd.io_http.cognitive import DictionaryExamples
    d = DictionaryExamples(textCol="t", translationCol="tr")
    row = pd.Series({"t": "hello", "tr": "hola"})
    p = d._payload(row)
    assert p == [{"Text": "hello", "Translation": "hola"}]


def test_azure_search_writer_batches(mock_service):
    """AzureSearchWriter (AzureSearchAPI.scala sink): docs batched into
    @search.action-tagged POSTs; every row carries the batch response."""
    from mmlspark_amd.io_http.cognitive import AzureSearchWriter
    docs = [{"id": str(i), "text": f"d{i}"} for i in range(7)]
    df = pd.DataFrame({"doc": docs})
    w = AzureSearchWriter(url=mock_service, batchSize=3, actionType="upload",
                          subscriptionKey="k", outputCol="result")
    out = w.transform(df)
    assert len(out) == 7
    assert out["result"].notna().all()
    # 7 docs at batchSize 3 → 3 distinct batch responses (dict per row)
    uniq = {json.dumps(r, sort_keys=True, default=str)
            for r in out["result"]}
    assert len(uniq) <= 3


def test_speech_to_text_posts_audio_bytes():
    from mmlspark_amd.io_http.cognitive import SpeechToTextSDK
    s = SpeechToTextSDK(url="http://x/", audioBytesCol="audio")
    row = pd.Series({"audio": b"RIFFxxxx"})
    assert s._payload(row) is None           # body is raw audio, not JSON
    assert s._headers(row)["Content-Type"] == "audio/wav"


def test_unroll_and_resize_stages():
    """image/ subpackage parity (UnrollImage.scala:151,186 +
    ResizeImageTransformer.scala): unroll to [0,1] floats, binary decode
    path, standalone resize."""
    from mmlspark_amd.io_http.files import encode_image
    from mmlspark_amd.models.images import (ResizeImageTransformer,
                                            UnrollBinaryImage, UnrollImage)
    rng = np.random.default_rng(4)
    img = rng.integers(0, 255, size=(20, 16, 3)).astype(np.uint8)
    df = pd.DataFrame({"image": [img], "data": [encode_image(img, "png")]})
    r = ResizeImageTransformer(height=8, width=8).transform(df)
    assert np.asarray(r["image"].iloc[0]).shape == (8, 8, 3)
    u = UnrollImage().transform(df)["unrolled"].iloc[0]
    assert u.shape == (20 * 16 * 3,) and u.max() > 1  # raw 0-255, CHW
    # channel-major: first W entries are channel 0 of row 0
    np.testing.assert_allclose(u[:16], img[0, :, 0].astype(np.float64))
    ub = UnrollBinaryImage(height=8, width=8).transform(df)["unrolled"].iloc[0]
    assert ub.shape == (8 * 8 * 3,)


def test_image_transformer_batched_matches_per_image():
    """The uniform-shape batched path must match the per-image path for
    every op (resize/crop/flip/gray/blur/gaussian/threshold/normalize)."""
    from mmlspark_amd.models.images import ImageTransformer
    rng = np.random.default_rng(7)
    imgs = [rng.integers(0, 255, size=(48, 40, 3)).astype(np.uint8)
            for _ in range(6)]
    mixed = imgs[:5] + [rng.integers(0, 255, size=(32, 32, 3)).astype(np.uint8)]

    pipelines = [
        lambda t: t.resize(24, 28).crop(2, 3, 16, 18).normalize(
            mean=[0.4, 0.5, 0.6], std=[0.2, 0.3, 0.4]),
        lambda t: t.flip(1).blur(3, 3).threshold(90.0, 255.0),
        lambda t: t.colorFormat("gray").gaussianKernel(5, 1.2),
        lambda t: t.colorFormat("bgr2rgb").flip(0).flip(-1),
    ]
    for make in pipelines:
        t = make(ImageTransformer(inputCol="image", outputCol="o"))
        batched = t.transform(pd.DataFrame({"image": imgs}))["o"]
        # mixed shapes force the per-image fallback on the SAME transformer
        per_img = t.transform(pd.DataFrame({"image": mixed}))["o"]
        for i in range(5):
            np.testing.assert_allclose(
                np.asarray(batched.iloc[i], dtype=np.float64),
                np.asarray(per_img.iloc[i], dtype=np.float64),
                atol=1e-3)


def test_image_normalize_keeps_negative_values():
    """normalize outputs are signed; they must not be clamped at 0."""
    from mmlspark_amd.models.images import ImageTransformer
    img = np.zeros((8, 8, 3), dtype=np.uint8)  # → (0/255 - mean)/std < 0
    t = ImageTransformer(inputCol="image", outputCol="o").normalize(
        mean=[0.485, 0.456, 0.406], std=[0.229, 0.224, 0.225])
    for frame in (pd.DataFrame({"image": [img]}),              # per-image
                  pd.DataFrame({"image": [img, img]})):        # batched
        out = np.asarray(t.transform(frame)["o"].iloc[0])
        assert out.min() < -1.0, out.min()
