"""Text featurization (core/featurize/text parity): TextFeaturizer
(tokenize → ngrams → TF(-IDF)), MultiNGram, PageSplitter."""
from __future__ import annotations

import re
from typing import List

import numpy as np

from ..core.param import Param, toBool, toInt, toList, toString
from ..core.pipeline import Estimator, Model, Transformer
from ..core.registry import register
from ..core.schema import SparseVector
from ..models.vw.murmur import hash_string


def _tokenize(s: str, pattern: str = r"\s+", to_lower: bool = True,
              min_len: int = 1) -> List[str]:
    if to_lower:
        s = s.lower()
    toks = [t for t in re.split(pattern, str(s)) if len(t) >= min_len]
    return toks


def _ngrams(tokens: List[str], n: int) -> List[str]:
    if n <= 1:
        return list(tokens)
    return [" ".join(tokens[i:i + n]) for i in range(len(tokens) - n + 1)]


@register
class TextFeaturizer(Estimator):
    """tokenize → ngram → hashed TF → IDF (TextFeaturizer.scala:196)."""
    inputCol = Param("inputCol", "text column", "text")
    outputCol = Param("outputCol", "feature vector column", "features")
    useTokenizer = Param("useTokenizer", "tokenize input", True, toBool)
    tokenizerPattern = Param("tokenizerPattern", "split regex", r"\s+", toString)
    toLowercase = Param("toLowercase", "lowercase", True, toBool)
    minTokenLength = Param("minTokenLength", "min token length", 0, toInt)
    useNGram = Param("useNGram", "add ngrams", False, toBool)
    nGramLength = Param("nGramLength", "ngram order", 2, toInt)
    numFeatures = Param("numFeatures", "hash dimension", 1 << 18, toInt)
    useIDF = Param("useIDF", "apply inverse document frequency", True, toBool)
    minDocFreq = Param("minDocFreq", "min document frequency", 1, toInt)
    useStopWordsRemover = Param("useStopWordsRemover", "drop stop words "
                                "(TextFeaturizer.scala useStopWordsRemover)",
                                False, toBool)
    stopWords = Param("stopWords", "stop words (comma-separated string or "
                      "list); default English list when unset", None)
    caseSensitiveStopWords = Param("caseSensitiveStopWords",
                                   "match stop words case-sensitively",
                                   False, toBool)
    binary = Param("binary", "binary term frequency (presence instead of "
                   "counts)", False, toBool)

    _DEFAULT_STOP_WORDS = (
        "a,an,the,and,or,but,if,then,is,are,was,were,be,been,of,to,in,on,"
        "for,with,as,at,by,it,its,this,that,these,those,from,not,no,so,too,"
        "very,can,will,just,do,does,did,have,has,had,i,you,he,she,we,they")

    def _stop_set(self):
        cached = getattr(self, "_stop_cache", None)
        if cached is not None:
            return cached
        sw = self.get("stopWords")
        if sw is None:
            words = self._DEFAULT_STOP_WORDS.split(",")
        elif isinstance(sw, str):
            words = sw.split(",")
        else:
            words = list(sw)
        if not self.get("caseSensitiveStopWords"):
            words = [w.lower() for w in words]
        self._stop_cache = set(words)
        return self._stop_cache

    def _terms(self, s):
        toks = _tokenize(s, self.get("tokenizerPattern"),
                         self.get("toLowercase"),
                         self.get("minTokenLength")) \
            if self.get("useTokenizer") else list(s)
        if self.get("useStopWordsRemover"):
            stop = self._stop_set()
            cs = self.get("caseSensitiveStopWords")
            toks = [t for t in toks if (t if cs else t.lower()) not in stop]
        if self.get("useNGram"):
            toks = toks + _ngrams(toks, self.get("nGramLength"))
        return toks

    def _fit(self, df):
        dim = self.get("numFeatures")
        n_docs = len(df)
        df_counts = np.zeros(dim, dtype=np.int64)
        for s in df[self.get("inputCol")]:
            seen = {hash_string(t) % dim for t in self._terms(s)}
            for h in seen:
                df_counts[h] += 1
        idf = np.zeros(dim, dtype=np.float32)
        if self.get("useIDF"):
            mask = df_counts >= self.get("minDocFreq")
            idf[mask] = np.log((n_docs + 1) / (df_counts[mask] + 1.0))
        m = TextFeaturizerModel()
        m.set("idf", idf if self.get("useIDF") else np.ones(dim, np.float32))
        for p in ("inputCol", "outputCol", "useTokenizer", "tokenizerPattern",
                  "toLowercase", "minTokenLength", "useNGram", "nGramLength",
                  "numFeatures", "useStopWordsRemover", "stopWords",
                  "caseSensitiveStopWords", "binary"):
            m.set(p, self.get(p))
        return m


@register
class TextFeaturizerModel(Model):
    inputCol = Param("inputCol", "text column", "text")
    outputCol = Param("outputCol", "feature vector column", "features")
    useTokenizer = Param("useTokenizer", "tokenize input", True, toBool)
    tokenizerPattern = Param("tokenizerPattern", "split regex", r"\s+", toString)
    toLowercase = Param("toLowercase", "lowercase", True, toBool)
    minTokenLength = Param("minTokenLength", "min token length", 0, toInt)
    useNGram = Param("useNGram", "add ngrams", False, toBool)
    nGramLength = Param("nGramLength", "ngram order", 2, toInt)
    numFeatures = Param("numFeatures", "hash dimension", 1 << 18, toInt)
    useStopWordsRemover = Param("useStopWordsRemover", "drop stop words",
                                False, toBool)
    stopWords = Param("stopWords", "stop words", None)
    caseSensitiveStopWords = Param("caseSensitiveStopWords",
                                   "case-sensitive match", False, toBool)
    binary = Param("binary", "binary term frequency", False, toBool)
    idf = Param("idf", "idf weights", None, is_complex=True)

    def _transform(self, df):
        dim = self.get("numFeatures")
        idf = np.asarray(self.get("idf"))
        fe = TextFeaturizer()
        for p in ("useTokenizer", "tokenizerPattern", "toLowercase",
                  "minTokenLength", "useNGram", "nGramLength",
                  "useStopWordsRemover", "stopWords",
                  "caseSensitiveStopWords"):
            fe.set(p, self.get(p))
        vecs = []
        for s in df[self.get("inputCol")]:
            counts = {}
            for t in fe._terms(s):
                h = hash_string(t) % dim
                counts[h] = counts.get(h, 0) + 1
            idx = np.array(sorted(counts), dtype=np.int32)
            val = np.array([counts[i] for i in idx], dtype=np.float32)
            if self.get("binary"):
                val = np.ones_like(val)
            val = val * idf[idx]
            vecs.append(SparseVector(dim, idx, val))
        out = df.copy()
        out[self.get("outputCol")] = vecs
        return out


@register
class MultiNGram(Transformer):
    """Concatenate ngrams of several orders (MultiNGram.scala)."""
    inputCol = Param("inputCol", "token-list column", "tokens")
    outputCol = Param("outputCol", "ngram column", "ngrams")
    lengths = Param("lengths", "ngram orders", [1, 2, 3], toList)

    def _transform(self, df):
        out = df.copy()
        out[self.get("outputCol")] = [
            sum((_ngrams(list(toks), n) for n in self.get("lengths")), [])
            for toks in df[self.get("inputCol")]]
        return out


@register
class PageSplitter(Transformer):
    """Split documents into pages of bounded length (PageSplitter.scala)."""
    inputCol = Param("inputCol", "text column", "text")
    outputCol = Param("outputCol", "pages column", "pages")
    maximumPageLength = Param("maximumPageLength", "max chars/page", 5000, toInt)
    minimumPageLength = Param("minimumPageLength", "min chars/page", 4500, toInt)
    boundaryRegex = Param("boundaryRegex", "preferred split regex", r"\s", toString)

    def _transform(self, df):
        mx = self.get("maximumPageLength")
        mn = self.get("minimumPageLength")
        pat = re.compile(self.get("boundaryRegex"))

        def split(s):
            s = str(s)
            pages = []
            while len(s) > mx:
                cut = mx
                m = None
                for m_ in pat.finditer(s, mn, mx):
                    m = m_
                if m is not None:
                    cut = m.start() + 1
                pages.append(s[:cut])
                s = s[cut:]
            pages.append(s)
            return pages
        out = df.copy()
        out[self.get("outputCol")] = df[self.get("inputCol")].map(split)
        return out
