"""Pure-numpy PNG codec (decode + encode) — stdlib zlib only.

Covers the PNG subset that matters for ML pipelines: 8-bit grayscale /
gray+alpha / RGB / RGBA / palette, all five scanline filters, single
IDAT-stream non-interlaced images (interlaced Adam7 rejected with a clear
error).  The analog of the reference's ImageUtils decode/encode
(core/.../core/image/ImageUtils.scala) without any native codec library.
"""
from __future__ import annotations

import struct
import zlib

import numpy as np

_SIG = b"\x89PNG\r\n\x1a\n"
# channels per color type
_CHANNELS = {0: 1, 2: 3, 3: 1, 4: 2, 6: 4}


def _paeth(a, b, c):
    p = a.astype(np.int32) + b.astype(np.int32) - c.astype(np.int32)
    pa = np.abs(p - a)
    pb = np.abs(p - b)
    pc = np.abs(p - c)
    out = np.where((pa <= pb) & (pa <= pc), a, np.where(pb <= pc, b, c))
    return out.astype(np.uint8)


def decode_png(data: bytes) -> np.ndarray:
    if data[:8] != _SIG:
        raise ValueError("not a PNG")
    pos = 8
    width = height = bit_depth = color_type = interlace = None
    idat = bytearray()
    palette = None
    trns = None
    while pos + 8 <= len(data):
        (length,), ctype = struct.unpack(">I", data[pos:pos + 4]), \
            data[pos + 4:pos + 8]
        chunk = data[pos + 8:pos + 8 + length]
        pos += 12 + length
        if ctype == b"IHDR":
            width, height, bit_depth, color_type, _, _, interlace = \
                struct.unpack(">IIBBBBB", chunk)
        elif ctype == b"PLTE":
            palette = np.frombuffer(chunk, np.uint8).reshape(-1, 3)
        elif ctype == b"tRNS":
            trns = np.frombuffer(chunk, np.uint8)
        elif ctype == b"IDAT":
            idat.extend(chunk)
        elif ctype == b"IEND":
            break
    if width is None:
        raise ValueError("PNG missing IHDR")
    if bit_depth != 8:
        raise ValueError(f"PNG bit depth {bit_depth} unsupported (8 only)")
    if interlace:
        raise ValueError("interlaced (Adam7) PNG unsupported")
    ch = _CHANNELS.get(color_type)
    if ch is None:
        raise ValueError(f"PNG color type {color_type} unsupported")

    raw = zlib.decompress(bytes(idat))
    stride = width * ch
    expected = height * (stride + 1)
    if len(raw) < expected:
        raise ValueError("PNG data truncated")
    rows = np.frombuffer(raw[:expected], np.uint8).reshape(height, stride + 1)
    filters = rows[:, 0]
    img = np.zeros((height, stride), np.uint8)
    zero_row = np.zeros(stride, np.uint8)
    for y in range(height):
        cur = rows[y, 1:].copy()
        f = filters[y]
        prev = img[y - 1] if y else zero_row
        if f == 0:
            img[y] = cur
        elif f == 2:  # Up — vectorized
            img[y] = cur + prev
        elif f in (1, 3, 4):  # Sub / Average / Paeth need left-to-right
            left = np.zeros(ch, np.uint8)
            line = img[y]
            for x0 in range(0, stride, ch):
                a = line[x0 - ch:x0] if x0 else left
                b = prev[x0:x0 + ch]
                if f == 1:
                    line[x0:x0 + ch] = cur[x0:x0 + ch] + a
                elif f == 3:
                    line[x0:x0 + ch] = cur[x0:x0 + ch] + (
                        (a.astype(np.uint16) + b) // 2).astype(np.uint8)
                else:
                    c = prev[x0 - ch:x0] if x0 else left
                    line[x0:x0 + ch] = cur[x0:x0 + ch] + _paeth(a, b, c)
        else:
            raise ValueError(f"PNG filter {f} invalid")
    out = img.reshape(height, width, ch)
    if color_type == 3:  # palette
        if palette is None:
            raise ValueError("palette PNG without PLTE")
        idx = out[:, :, 0]
        rgb = palette[idx]
        if trns is not None:
            a = np.full(256, 255, np.uint8)
            a[: len(trns)] = trns
            return np.dstack([rgb, a[idx]])
        return rgb
    if ch == 1:
        return out[:, :, 0]
    return out


def encode_png(img: np.ndarray) -> bytes:
    img = np.asarray(img, np.uint8)
    if img.ndim == 2:
        img = img[:, :, None]
    h, w, ch = img.shape
    color_type = {1: 0, 2: 4, 3: 2, 4: 6}.get(ch)
    if color_type is None:
        raise ValueError(f"{ch}-channel image unsupported")
    raw = bytearray()
    for y in range(h):  # filter 0 every row (fast, zlib still compresses)
        raw.append(0)
        raw.extend(img[y].tobytes())

    def chunk(ctype: bytes, payload: bytes) -> bytes:
        return (struct.pack(">I", len(payload)) + ctype + payload
                + struct.pack(">I", zlib.crc32(ctype + payload) & 0xFFFFFFFF))

    ihdr = struct.pack(">IIBBBBB", w, h, 8, color_type, 0, 0, 0)
    return (_SIG + chunk(b"IHDR", ihdr)
            + chunk(b"IDAT", zlib.compress(bytes(raw), 6))
            + chunk(b"IEND", b""))
