"""File IO: binary-file reading and basic image codecs + PowerBI writer.

Parity: core/.../io/binary/BinaryFileFormat.scala:112 ((path, bytes) rows),
image/ImageUtils decode/encode (165), powerbi/PowerBIWriter (114, batched
REST push).  Offline image decoding covers BMP / PPM / PGM / NPY natively
(no OpenCV/PIL in the image); JPEG/PNG raise a clear error."""
from __future__ import annotations

import glob
import json
import os
import struct

import numpy as np
import pandas as pd

from ..core.param import Param, toBool, toInt
from ..core.pipeline import Transformer
from ..core.registry import register


def read_binary_files(pattern: str, recursive: bool = True,
                      path_col: str = "path",
                      bytes_col: str = "bytes") -> pd.DataFrame:
    """Glob files → DataFrame of (path, bytes) (BinaryFileFormat reader)."""
    paths = sorted(glob.glob(pattern, recursive=recursive))
    rows = []
    for p in paths:
        if os.path.isfile(p):
            with open(p, "rb") as f:
                rows.append({path_col: p, bytes_col: f.read()})
    return pd.DataFrame(rows, columns=[path_col, bytes_col])


def write_binary_files(df: pd.DataFrame, out_dir: str, path_col: str = "path",
                       bytes_col: str = "bytes"):
    os.makedirs(out_dir, exist_ok=True)
    for _, row in df.iterrows():
        name = os.path.basename(str(row[path_col]))
        with open(os.path.join(out_dir, name), "wb") as f:
            f.write(bytes(row[bytes_col]))


# ------------------------------------------------------------- image codecs
def decode_image(data: bytes) -> np.ndarray:
    """bytes → HWC uint8 array. BMP/PPM/PGM/NPY/PNG/JPEG supported offline.
    Malformed input raises ValueError (never a struct/parse internal error)."""
    try:
        if data[:2] == b"BM":
            return _decode_bmp(data)
        if data[:2] in (b"P6", b"P5", b"P3", b"P2"):
            return _decode_pnm(data)
    except ValueError:
        raise
    except Exception as e:  # truncated/garbled container
        raise ValueError(f"malformed image data: {e}") from e
    if data[:6] == b"\x93NUMPY":
        import io
        return np.load(io.BytesIO(data))
    if data[:2] == b"\xff\xd8":
        from .jpeg_codec import decode_jpeg
        return decode_jpeg(data)
    if data[:8] == b"\x89PNG\r\n\x1a\n":
        from .png_codec import decode_png
        return decode_png(data)
    raise ValueError("unrecognized image format")


def encode_image(img: np.ndarray, fmt: str = "ppm") -> bytes:
    img = np.asarray(img)
    if fmt in ("ppm", "pnm"):
        h, w = img.shape[:2]
        if img.ndim == 2 or img.shape[2] == 1:
            hdr = f"P5\n{w} {h}\n255\n".encode()
            return hdr + img.reshape(h, w).astype(np.uint8).tobytes()
        hdr = f"P6\n{w} {h}\n255\n".encode()
        return hdr + img[:, :, :3].astype(np.uint8).tobytes()
    if fmt == "npy":
        import io
        buf = io.BytesIO()
        np.save(buf, img)
        return buf.getvalue()
    if fmt == "png":
        from .png_codec import encode_png
        return encode_png(img)
    if fmt in ("jpg", "jpeg"):
        from .jpeg_codec import encode_jpeg
        return encode_jpeg(img)
    raise ValueError(f"unsupported encode format {fmt}")


def _decode_bmp(data: bytes) -> np.ndarray:
    off = struct.unpack_from("<I", data, 10)[0]
    hdr_size = struct.unpack_from("<I", data, 14)[0]
    w, h = struct.unpack_from("<ii", data, 18)
    bpp = struct.unpack_from("<H", data, 28)[0]
    if bpp not in (24, 32):
        raise ValueError(f"BMP bpp {bpp} unsupported")
    nch = bpp // 8
    flip = h > 0
    h = abs(h)
    row_size = ((w * nch + 3) // 4) * 4
    out = np.zeros((h, w, 3), dtype=np.uint8)
    for y in range(h):
        row = np.frombuffer(data, dtype=np.uint8, count=w * nch,
                            offset=off + y * row_size).reshape(w, nch)
        out[h - 1 - y if flip else y] = row[:, :3][:, ::-1]  # BGR→RGB
    return out


def _decode_pnm(data: bytes) -> np.ndarray:
    # parse header tokens (magic, width, height, [maxval])
    toks = []
    i = 2
    magic = data[:2]
    while len(toks) < (2 if magic in (b"P1", b"P4") else 3):
        while i < len(data) and data[i:i + 1].isspace():
            i += 1
        if data[i:i + 1] == b"#":
            while i < len(data) and data[i] != 0x0A:
                i += 1
            continue
        t = b""
        while i < len(data) and not data[i:i + 1].isspace():
            t += data[i:i + 1]
            i += 1
        toks.append(int(t))
    i += 1  # single whitespace after header
    w, h = toks[0], toks[1]
    if magic == b"P6":
        arr = np.frombuffer(data, np.uint8, w * h * 3, i).reshape(h, w, 3)
    elif magic == b"P5":
        arr = np.frombuffer(data, np.uint8, w * h, i).reshape(h, w, 1)
    else:  # ASCII P2/P3
        vals = np.array(data[i:].split(), dtype=np.int64)
        ch = 3 if magic == b"P3" else 1
        arr = vals.reshape(h, w, ch).astype(np.uint8)
    return np.ascontiguousarray(arr)


@register
class BinaryFileReader(Transformer):
    """Transformer shape over read_binary_files: input df has a path column."""
    pathCol = Param("pathCol", "path column", "path")
    bytesCol = Param("bytesCol", "output bytes column", "bytes")

    def _transform(self, df):
        out = df.copy()
        vals = []
        for p in df[self.get("pathCol")]:
            with open(p, "rb") as f:
                vals.append(f.read())
        out[self.get("bytesCol")] = vals
        return out


@register
class ImageReader(Transformer):
    """Decode a bytes column into image arrays (ImageUtils parity)."""
    bytesCol = Param("bytesCol", "bytes column", "bytes")
    imageCol = Param("imageCol", "output image column", "image")
    dropInvalid = Param("dropInvalid", "drop undecodable rows", False, toBool)

    def _transform(self, df):
        imgs, keep = [], []
        for b in df[self.get("bytesCol")]:
            try:
                imgs.append(decode_image(bytes(b)))
                keep.append(True)
            except ValueError:
                imgs.append(None)
                keep.append(False)
        out = df.copy()
        out[self.get("imageCol")] = imgs
        if self.get("dropInvalid"):
            out = out[pd.Series(keep, index=out.index)]
        return out


@register
class PowerBIWriter(Transformer):
    """Batched REST push of rows to a PowerBI-style endpoint
    (powerbi/PowerBIWriter.scala)."""
    url = Param("url", "push endpoint", "")
    batchSize = Param("batchSize", "rows per POST", 100, toInt)
    concurrency = Param("concurrency", "parallel posts", 1, toInt)

    def _transform(self, df):
        from .client import HTTPTransformer
        from .http_schema import HTTPRequestData
        bs = self.get("batchSize")
        reqs = []
        for s in range(0, len(df), bs):
            chunk = df.iloc[s:s + bs]
            body = json.dumps({"rows": chunk.to_dict("records")},
                              default=str).encode()
            reqs.append(HTTPRequestData(
                url=self.get("url"), method="POST",
                headers={"Content-Type": "application/json"}, entity=body))
        tmp = pd.DataFrame({"request": reqs})
        resp = HTTPTransformer(inputCol="request", outputCol="response",
                               concurrency=self.get("concurrency")).transform(tmp)
        codes = [r.statusCode for r in resp["response"]]
        bad = [c for c in codes if not (200 <= c < 300)]
        if bad:
            raise RuntimeError(f"PowerBI push failed for {len(bad)} batches: {bad[:5]}")
        return df
