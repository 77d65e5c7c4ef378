"""Pure-python baseline JPEG codec (stdlib + numpy/scipy only).

Decoder: baseline sequential DCT (SOF0), standard huffman tables from the
stream, 4:4:4 / 4:2:0 / 4:2:2 chroma subsampling, restart markers.
Encoder: baseline 4:4:4 with the Annex-K standard tables (quality-scaled).
8×8 FDCT/IDCT are the orthonormal scipy DCT-II/III, which match the JPEG
definitions exactly.  Completes the ImageUtils codec surface
(core/.../core/image/ImageUtils.scala) for the offline image — correctness
over speed (python huffman), intended for test assets and small images.
"""
from __future__ import annotations

import struct
from typing import Dict, List, Tuple

import numpy as np
from scipy.fft import dctn, idctn

ZIGZAG = np.array([
    0, 1, 8, 16, 9, 2, 3, 10, 17, 24, 32, 25, 18, 11, 4, 5,
    12, 19, 26, 33, 40, 48, 41, 34, 27, 20, 13, 6, 7, 14, 21, 28,
    35, 42, 49, 56, 57, 50, 43, 36, 29, 22, 15, 23, 30, 37, 44, 51,
    58, 59, 52, 45, 38, 31, 39, 46, 53, 60, 61, 54, 47, 55, 62, 63])

# Annex K quantization tables (luminance, chrominance)
QL = np.array([
    16, 11, 10, 16, 24, 40, 51, 61, 12, 12, 14, 19, 26, 58, 60, 55,
    14, 13, 16, 24, 40, 57, 69, 56, 14, 17, 22, 29, 51, 87, 80, 62,
    18, 22, 37, 56, 68, 109, 103, 77, 24, 35, 55, 64, 81, 104, 113, 92,
    49, 64, 78, 87, 103, 121, 120, 101, 72, 92, 95, 98, 112, 100, 103, 99])
QC = np.array([
    17, 18, 24, 47, 99, 99, 99, 99, 18, 21, 26, 66, 99, 99, 99, 99,
    24, 26, 56, 99, 99, 99, 99, 99, 47, 66, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99,
    99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99, 99])

# Annex K huffman tables: (bits[1..16], values)
HT_DC_L = ([0, 1, 5, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0, 0, 0],
           list(range(12)))
HT_DC_C = ([0, 3, 1, 1, 1, 1, 1, 1, 1, 1, 1, 0, 0, 0, 0, 0],
           list(range(12)))
HT_AC_L = ([0, 2, 1, 3, 3, 2, 4, 3, 5, 5, 4, 4, 0, 0, 1, 0x7d], [
    0x01, 0x02, 0x03, 0x00, 0x04, 0x11, 0x05, 0x12, 0x21, 0x31, 0x41, 0x06,
    0x13, 0x51, 0x61, 0x07, 0x22, 0x71, 0x14, 0x32, 0x81, 0x91, 0xa1, 0x08,
    0x23, 0x42, 0xb1, 0xc1, 0x15, 0x52, 0xd1, 0xf0, 0x24, 0x33, 0x62, 0x72,
    0x82, 0x09, 0x0a, 0x16, 0x17, 0x18, 0x19, 0x1a, 0x25, 0x26, 0x27, 0x28,
    0x29, 0x2a, 0x34, 0x35, 0x36, 0x37, 0x38, 0x39, 0x3a, 0x43, 0x44, 0x45,
    0x46, 0x47, 0x48, 0x49, 0x4a, 0x53, 0x54, 0x55, 0x56, 0x57, 0x58, 0x59,
    0x5a, 0x63, 0x64, 0x65, 0x66, 0x67, 0x68, 0x69, 0x6a, 0x73, 0x74, 0x75,
    0x76, 0x77, 0x78, 0x79, 0x7a, 0x83, 0x84, 0x85, 0x86, 0x87, 0x88, 0x89,
    0x8a, 0x92, 0x93, 0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9a, 0xa2, 0xa3,
    0xa4, 0xa5, 0xa6, 0xa7, 0xa8, 0xa9, 0xaa, 0xb2, 0xb3, 0xb4, 0xb5, 0xb6,
    0xb7, 0xb8, 0xb9, 0xba, 0xc2, 0xc3, 0xc4, 0xc5, 0xc6, 0xc7, 0xc8, 0xc9,
    0xca, 0xd2, 0xd3, 0xd4, 0xd5, 0xd6, 0xd7, 0xd8, 0xd9, 0xda, 0xe1, 0xe2,
    0xe3, 0xe4, 0xe5, 0xe6, 0xe7, 0xe8, 0xe9, 0xea, 0xf1, 0xf2, 0xf3, 0xf4,
    0xf5, 0xf6, 0xf7, 0xf8, 0xf9, 0xfa])
HT_AC_C = ([0, 2, 1, 2, 4, 4, 3, 4, 7, 5, 4, 4, 0, 1, 2, 0x77], [
    0x00, 0x01, 0x02, 0x03, 0x11, 0x04, 0x05, 0x21, 0x31, 0x06, 0x12, 0x41,
    0x51, 0x07, 0x61, 0x71, 0x13, 0x22, 0x32, 0x81, 0x08, 0x14, 0x42, 0x91,
    0xa1, 0xb1, 0xc1, 0x09, 0x23, 0x33, 0x52, 0xf0, 0x15, 0x62, 0x72, 0xd1,
    0x0a, 0x16, 0x24, 0x34, 0xe1, 0x25, 0xf1, 0x17, 0x18, 0x19, 0x1a, 0x26,
    0x27, 0x28, 0x29, 0x2a, 0x35, 0x36, 0x37, 0x38, 0x39, 0x3a, 0x43, 0x44,
    0x45, 0x46, 0x47, 0x48, 0x49, 0x4a, 0x53, 0x54, 0x55, 0x56, 0x57, 0x58,
    0x59, 0x5a, 0x63, 0x64, 0x65, 0x66, 0x67, 0x68, 0x69, 0x6a, 0x73, 0x74,
    0x75, 0x76, 0x77, 0x78, 0x79, 0x7a, 0x82, 0x83, 0x84, 0x85, 0x86, 0x87,
    0x88, 0x89, 0x8a, 0x92, 0x93, 0x94, 0x95, 0x96, 0x97, 0x98, 0x99, 0x9a,
    0xa2, 0xa3, 0xa4, 0xa5, 0xa6, 0xa7, 0xa8, 0xa9, 0xaa, 0xb2, 0xb3, 0xb4,
    0xb5, 0xb6, 0xb7, 0xb8, 0xb9, 0xba, 0xc2, 0xc3, 0xc4, 0xc5, 0xc6, 0xc7,
    0xc8, 0xc9, 0xca, 0xd2, 0xd3, 0xd4, 0xd5, 0xd6, 0xd7, 0xd8, 0xd9, 0xda,
    0xe2, 0xe3, 0xe4, 0xe5, 0xe6, 0xe7, 0xe8, 0xe9, 0xea, 0xf2, 0xf3, 0xf4,
    0xf5, 0xf6, 0xf7, 0xf8, 0xf9, 0xfa])


def _build_codes(bits: List[int], values: List[int]) -> Dict[int, Tuple[int, int]]:
    """value → (code, length)."""
    codes = {}
    code = 0
    k = 0
    for length in range(1, 17):
        for _ in range(bits[length - 1]):
            codes[values[k]] = (code, length)
            code += 1
            k += 1
        code <<= 1
    return codes


def _build_decoder(bits: List[int], values: List[int]) -> Dict[Tuple[int, int], int]:
    """(length, code) → value."""
    table = {}
    code = 0
    k = 0
    for length in range(1, 17):
        for _ in range(bits[length - 1]):
            table[(length, code)] = values[k]
            code += 1
            k += 1
        code <<= 1
    return table


class _BitWriter:
    def __init__(self):
        self.out = bytearray()
        self.acc = 0
        self.nbits = 0

    def write(self, code: int, length: int):
        self.acc = (self.acc << length) | (code & ((1 << length) - 1))
        self.nbits += length
        while self.nbits >= 8:
            b = (self.acc >> (self.nbits - 8)) & 0xFF
            self.out.append(b)
            if b == 0xFF:
                self.out.append(0x00)  # byte stuffing
            self.nbits -= 8

    def flush(self):
        if self.nbits:
            pad = 8 - self.nbits
            self.write((1 << pad) - 1, pad)


class _BitReader:
    """Lazy bit reader: one byte at a time, FF00 de-stuffing, markers read
    as zero bits (the scan decoder stops via block counts / RST skips)."""

    def __init__(self, data: bytes):
        self.data = data
        self.pos = 0
        self.cur = 0
        self.nbits = 0

    def read_bit(self) -> int:
        if self.nbits == 0:
            if self.pos >= len(self.data):
                return 0
            b = self.data[self.pos]
            if b == 0xFF:
                nxt = self.data[self.pos + 1] if self.pos + 1 < len(self.data) else 0xD9
                if nxt == 0x00:
                    self.pos += 2  # stuffed FF
                else:
                    return 0  # at a marker: emit zeros, do not consume
            else:
                self.pos += 1
            self.cur = b
            self.nbits = 8
        self.nbits -= 1
        return (self.cur >> self.nbits) & 1

    def read_bits(self, n: int) -> int:
        v = 0
        for _ in range(n):
            v = (v << 1) | self.read_bit()
        return v

    def align_and_skip_rst(self):
        self.nbits = 0  # drop partial byte
        if self.pos + 1 < len(self.data) and self.data[self.pos] == 0xFF \
                and 0xD0 <= self.data[self.pos + 1] <= 0xD7:
            self.pos += 2


def _decode_huff(br: _BitReader, table: Dict[Tuple[int, int], int]) -> int:
    code = 0
    for length in range(1, 17):
        code = (code << 1) | br.read_bit()
        v = table.get((length, code))
        if v is not None:
            return v
    raise ValueError("invalid huffman code")


def _extend(v: int, n: int) -> int:
    if n == 0:
        return 0
    return v if v >= (1 << (n - 1)) else v - (1 << n) + 1


def decode_jpeg(data: bytes) -> np.ndarray:
    """Decode a JPEG: the native C++ decoder (baseline + progressive SOF2,
    jpeg_native.cpp) when built, else the pure-Python baseline path below."""
    try:
        import torch  # noqa: F401 — loads libc10 for the extension
        from . import _jpeg_native
    except ImportError:
        return _decode_jpeg_py(data)
    try:
        return _jpeg_native.decode_jpeg(bytes(data)).numpy()
    except RuntimeError as e:  # same error contract as the Python decoder
        raise ValueError(str(e)) from e


def _decode_jpeg_py(data: bytes) -> np.ndarray:
    if data[:2] != b"\xff\xd8":
        raise ValueError("not a JPEG")
    pos = 2
    qt: Dict[int, np.ndarray] = {}
    huff_dc: Dict[int, dict] = {}
    huff_ac: Dict[int, dict] = {}
    frame = None
    restart_interval = 0
    while pos + 4 <= len(data):
        if data[pos] != 0xFF:
            pos += 1
            continue
        marker = data[pos + 1]
        pos += 2
        if marker in (0xD8, 0x01) or 0xD0 <= marker <= 0xD7:
            continue
        (seglen,) = struct.unpack(">H", data[pos:pos + 2])
        seg = data[pos + 2:pos + seglen]
        if marker == 0xDB:  # DQT
            o = 0
            while o < len(seg):
                pq, tq = seg[o] >> 4, seg[o] & 15
                o += 1
                if pq:
                    tbl = np.frombuffer(seg[o:o + 128], ">u2").astype(np.int32)
                    o += 128
                else:
                    tbl = np.frombuffer(seg[o:o + 64], np.uint8).astype(np.int32)
                    o += 64
                qt[tq] = tbl
        elif marker == 0xC4:  # DHT
            o = 0
            while o < len(seg):
                tc, th = seg[o] >> 4, seg[o] & 15
                bits = list(seg[o + 1:o + 17])
                n = sum(bits)
                vals = list(seg[o + 17:o + 17 + n])
                (huff_ac if tc else huff_dc)[th] = _build_decoder(bits, vals)
                o += 17 + n
        elif marker == 0xC0:  # SOF0 baseline
            prec, h, w, nc = seg[0], *struct.unpack(">HH", seg[1:5]), seg[5]
            comps = []
            for i in range(nc):
                cid, hv, tq = seg[6 + 3 * i:9 + 3 * i]
                comps.append({"id": cid, "h": hv >> 4, "v": hv & 15, "tq": tq})
            frame = {"h": h, "w": w, "comps": comps}
        elif marker in (0xC1, 0xC2, 0xC3):
            raise ValueError("only baseline JPEG (SOF0) is supported")
        elif marker == 0xDD:  # DRI
            (restart_interval,) = struct.unpack(">H", seg[:2])
        elif marker == 0xDA:  # SOS
            ns = seg[0]
            scan = []
            for i in range(ns):
                cs, tda = seg[1 + 2 * i], seg[2 + 2 * i]
                scan.append({"cs": cs, "td": tda >> 4, "ta": tda & 15})
            pos += seglen
            return _decode_scan(data, pos, frame, scan, qt, huff_dc, huff_ac,
                                restart_interval)
        pos += seglen
    raise ValueError("no SOS marker found")


def _decode_scan(data, pos, frame, scan, qt, huff_dc, huff_ac, dri):
    h, w, comps = frame["h"], frame["w"], frame["comps"]
    hmax = max(c["h"] for c in comps)
    vmax = max(c["v"] for c in comps)
    mcux = (w + 8 * hmax - 1) // (8 * hmax)
    mcuy = (h + 8 * vmax - 1) // (8 * vmax)
    planes = []
    for c in comps:
        planes.append(np.zeros((mcuy * c["v"] * 8, mcux * c["h"] * 8),
                               np.float32))
    by_id = {s["cs"]: s for s in scan}
    br = _BitReader(data[pos:])
    pred = [0] * len(comps)
    mcu_count = 0
    for my in range(mcuy):
        for mx in range(mcux):
            if dri and mcu_count and mcu_count % dri == 0:
                br.align_and_skip_rst()
                pred = [0] * len(comps)
            for ci, c in enumerate(comps):
                s = by_id[c["id"]]
                for vy in range(c["v"]):
                    for vx in range(c["h"]):
                        blk = np.zeros(64, np.float32)
                        t = _decode_huff(br, huff_dc[s["td"]])
                        diff = _extend(br.read_bits(t), t)
                        pred[ci] += diff
                        blk[0] = pred[ci]
                        k = 1
                        while k < 64:
                            rs = _decode_huff(br, huff_ac[s["ta"]])
                            r, sz = rs >> 4, rs & 15
                            if rs == 0x00:
                                break
                            if rs == 0xF0:
                                k += 16
                                continue
                            k += r
                            if k > 63:
                                break
                            blk[k] = _extend(br.read_bits(sz), sz)
                            k += 1
                        blk = blk * qt[c["tq"]]
                        sq = np.zeros(64, np.float32)
                        sq[ZIGZAG] = blk
                        px = idctn(sq.reshape(8, 8), norm="ortho") + 128.0
                        y0 = (my * c["v"] + vy) * 8
                        x0 = (mx * c["h"] + vx) * 8
                        planes[ci][y0:y0 + 8, x0:x0 + 8] = px
            mcu_count += 1
    # upsample to full size
    out = []
    for c, p in zip(comps, planes):
        ry, rx = vmax // c["v"], hmax // c["h"]
        if ry > 1 or rx > 1:
            p = np.repeat(np.repeat(p, ry, axis=0), rx, axis=1)
        out.append(p[:h, :w])
    if len(out) == 1:
        return np.clip(out[0] + 0.5, 0, 255).astype(np.uint8)
    Y, Cb, Cr = out[0], out[1] - 128.0, out[2] - 128.0
    r = Y + 1.402 * Cr
    g = Y - 0.344136 * Cb - 0.714136 * Cr
    b = Y + 1.772 * Cb
    return np.clip(np.dstack([r, g, b]) + 0.5, 0, 255).astype(np.uint8)


def _quality_tables(quality: int):
    quality = min(100, max(1, quality))
    scale = 5000 // quality if quality < 50 else 200 - 2 * quality
    ql = np.clip((QL * scale + 50) // 100, 1, 255).astype(np.int32)
    qc = np.clip((QC * scale + 50) // 100, 1, 255).astype(np.int32)
    return ql, qc


def _mag(v: int) -> int:
    return int(v).bit_length() if v >= 0 else int(-v).bit_length()


def encode_jpeg(img: np.ndarray, quality: int = 90) -> bytes:
    img = np.asarray(img, np.uint8)
    gray = img.ndim == 2 or img.shape[2] == 1
    if gray:
        planes = [img.reshape(img.shape[0], img.shape[1]).astype(np.float32)]
    else:
        rgb = img[:, :, :3].astype(np.float32)
        r, g, b = rgb[:, :, 0], rgb[:, :, 1], rgb[:, :, 2]
        planes = [0.299 * r + 0.587 * g + 0.114 * b,
                  -0.168736 * r - 0.331264 * g + 0.5 * b + 128.0,
                  0.5 * r - 0.418688 * g - 0.081312 * b + 128.0]
    h, w = planes[0].shape
    ql, qc = _quality_tables(quality)
    qts = [ql] + ([qc, qc] if not gray else [])
    dc_codes = [_build_codes(*HT_DC_L), _build_codes(*HT_DC_C)]
    ac_codes = [_build_codes(*HT_AC_L), _build_codes(*HT_AC_C)]

    bw = _BitWriter()
    pred = [0] * len(planes)
    bh, bwid = (h + 7) // 8, (w + 7) // 8
    padded = []
    for p in planes:
        pp = np.empty((bh * 8, bwid * 8), np.float32)
        pp[:h, :w] = p
        pp[h:, :w] = p[-1:, :]
        pp[:, w:] = pp[:, w - 1:w]
        padded.append(pp)
    for by in range(bh):
        for bx in range(bwid):
            for ci, p in enumerate(padded):
                tsel = 0 if ci == 0 else 1
                q = ql if ci == 0 else qc
                block = p[by * 8:by * 8 + 8, bx * 8:bx * 8 + 8] - 128.0
                coef = dctn(block, norm="ortho")
                # zigzag-order coefficients ÷ zigzag-order quant steps
                zz = np.rint(coef.reshape(-1)[ZIGZAG]
                             / q.reshape(-1)[ZIGZAG]).astype(np.int64)
                diff = int(zz[0]) - pred[ci]
                pred[ci] = int(zz[0])
                n = _mag(diff)
                code, ln = dc_codes[tsel][n]
                bw.write(code, ln)
                if n:
                    bw.write(diff if diff >= 0 else diff + (1 << n) - 1, n)
                run = 0
                last_nz = 0
                for k in range(1, 64):
                    if zz[k]:
                        last_nz = k
                for k in range(1, last_nz + 1):
                    v = int(zz[k])
                    if v == 0:
                        run += 1
                        continue
                    while run > 15:
                        code, ln = ac_codes[tsel][0xF0]
                        bw.write(code, ln)
                        run -= 16
                    n = _mag(v)
                    code, ln = ac_codes[tsel][(run << 4) | n]
                    bw.write(code, ln)
                    bw.write(v if v >= 0 else v + (1 << n) - 1, n)
                    run = 0
                if last_nz < 63:
                    code, ln = ac_codes[tsel][0x00]
                    bw.write(code, ln)
    bw.flush()

    def seg(marker: int, payload: bytes) -> bytes:
        return bytes([0xFF, marker]) + struct.pack(">H", len(payload) + 2) \
            + payload

    out = bytearray(b"\xff\xd8")
    out += seg(0xDB, bytes([0x00]) + bytes(int(x) for x in ql[ZIGZAG]))
    if not gray:
        out += seg(0xDB, bytes([0x01]) + bytes(int(x) for x in qc[ZIGZAG]))
    nc = 1 if gray else 3
    sof = struct.pack(">BHHB", 8, h, w, nc)
    for i in range(nc):
        sof += bytes([i + 1, 0x11, 0 if i == 0 else 1])
    out += seg(0xC0, sof)
    for tc, tsel, (bits, vals) in ((0, 0, HT_DC_L), (1, 0, HT_AC_L),
                                   (0, 1, HT_DC_C), (1, 1, HT_AC_C)):
        if gray and tsel == 1:
            continue
        out += seg(0xC4, bytes([(tc << 4) | tsel]) + bytes(bits)
                   + bytes(vals))
    sos = bytes([nc])
    for i in range(nc):
        sos += bytes([i + 1, 0x00 if i == 0 else 0x11])
    sos += bytes([0, 63, 0])
    out += seg(0xDA, sos)
    out += bw.out
    out += b"\xff\xd9"
    return bytes(out)
