"""File IO codecs, PowerBI writer, codegen stub/doc generation."""
import os

import json

import numpy as np
import pandas as pd
import pytest

from mmlspark_amd.io_http.files import (BinaryFileReader, ImageReader,
                                        PowerBIWriter, decode_image,
                                        encode_image, read_binary_files,
                                        write_binary_files)


def test_binary_file_roundtrip(tmp_path):
    for i in range(3):
        with open(tmp_path / f"f{i}.bin", "wb") as f:
            f.write(bytes([i] * 10))
    df = read_binary_files(str(tmp_path / "*.bin"))
    assert len(df) == 3
    assert df["bytes"].iloc[1] == bytes([1] * 10)
    out_dir = tmp_path / "out"
    write_binary_files(df, str(out_dir))
    assert sorted(os.listdir(out_dir)) == ["f0.bin", "f1.bin", "f2.bin"]
    rdr = BinaryFileReader()
    df2 = rdr.transform(pd.DataFrame({"path": [str(tmp_path / "f0.bin")]}))
    assert df2["bytes"].iloc[0] == bytes([0] * 10)


def test_image_codecs_roundtrip():
    rng = np.random.default_rng(0)
    img = rng.integers(0, 255, size=(9, 7, 3)).astype(np.uint8)
    data = encode_image(img, "ppm")
    back = decode_image(data)
    np.testing.assert_array_equal(img, back)
    gray = img[:, :, 0]
    back_g = decode_image(encode_image(gray, "ppm"))
    np.testing.assert_array_equal(gray, back_g[:, :, 0])
    back_npy = decode_image(encode_image(img, "npy"))
    np.testing.assert_array_equal(img, back_npy)
    with pytest.raises(ValueError):
        decode_image(b"\xff\xd8 fake jpeg")


def test_image_reader_transformer():
    img = np.zeros((4, 4, 3), dtype=np.uint8)
    df = pd.DataFrame({"bytes": [encode_image(img, "ppm"), b"garbage!"]})
    out = ImageReader(dropInvalid=True).transform(df)
    assert len(out) == 1
    assert out["image"].iloc[0].shape == (4, 4, 3)


def test_powerbi_writer_against_mock():
    from mmlspark_amd.serving.server import ServingServer
    received = []

    def handler(payloads):
        received.extend(payloads)
        return [{"ok": True} for _ in payloads]

    srv = ServingServer(handler, port=0, mode="continuous").start()
    try:
        df = pd.DataFrame({"a": range(7), "b": ["x"] * 7})
        w = PowerBIWriter(url=f"http://127.0.0.1:{srv.port}/", batchSize=3)
        w.transform(df)
        assert sum(len(p["rows"]) for p in received) == 7
    finally:
        srv.stop()


def test_codegen_outputs(tmp_path):
    from mmlspark_amd.core.codegen import (generate_docs, generate_r_wrappers,
                                           generate_stubs)
    n = generate_stubs(str(tmp_path / "stubs"))
    assert n > 40
    stub_files = os.listdir(tmp_path / "stubs")
    assert any("gbdt" in f for f in stub_files)
    content = open(tmp_path / "stubs" /
                   [f for f in stub_files if "gbdt" in f][0]).read()
    assert "class LightGBMClassifier" in content
    assert "def setNumLeaves(self, value: int)" in content

    nd = generate_docs(str(tmp_path / "docs"))
    assert nd == n
    assert os.path.exists(tmp_path / "docs" / "LightGBMClassifier.md")

    nr = generate_r_wrappers(str(tmp_path / "r" / "bindings.R"))
    assert nr == n
    rcode = open(tmp_path / "r" / "bindings.R").read()
    assert "ml_light_gbm_classifier" in rcode


def test_png_codec_roundtrip_and_filters():
    """Pure-numpy PNG codec (ImageUtils analog): encode→decode round trip
    for gray/RGB/RGBA, all five scanline filters, and palette images."""
    import struct
    import zlib
    from mmlspark_amd.io_http.files import decode_image, encode_image
    from mmlspark_amd.io_http.png_codec import decode_png

    rng = np.random.default_rng(0)
    for shape in [(17, 23), (16, 16, 3), (9, 5, 4)]:
        img = rng.integers(0, 256, size=shape).astype(np.uint8)
        back = decode_image(encode_image(img, "png"))
        assert back.shape == img.shape and (back == img).all()

    # decode with every filter type (hand-encoded rows 0..4)
    w, h = 8, 5
    img = rng.integers(0, 256, size=(h, w, 3)).astype(np.uint8)
    raws = bytearray()
    prev = [0] * (w * 3)
    for y, f in enumerate([0, 1, 2, 3, 4]):
        line = [int(v) for v in img[y].reshape(-1)]
        enc = []
        for x in range(w * 3):
            a = line[x - 3] if x >= 3 else 0
            b = prev[x]
            c = prev[x - 3] if x >= 3 else 0
            if f == 0:
                enc.append(line[x])
            elif f == 1:
                enc.append((line[x] - a) & 255)
            elif f == 2:
                enc.append((line[x] - b) & 255)
            elif f == 3:
                enc.append((line[x] - (a + b) // 2) & 255)
            else:
                p = a + b - c
                pa, pb, pc = abs(p - a), abs(p - b), abs(p - c)
                pr = a if pa <= pb and pa <= pc else (b if pb <= pc else c)
                enc.append((line[x] - pr) & 255)
        raws.append(f)
        raws.extend(bytes(enc))
        prev = line

    def chunk(t, p):
        return (struct.pack(">I", len(p)) + t + p
                + struct.pack(">I", zlib.crc32(t + p) & 0xFFFFFFFF))

    png = (b"\x89PNG\r\n\x1a\n"
           + chunk(b"IHDR", struct.pack(">IIBBBBB", w, h, 8, 2, 0, 0, 0))
           + chunk(b"IDAT", zlib.compress(bytes(raws)))
           + chunk(b"IEND", b""))
    assert (decode_png(png) == img).all()

    # palette (color type 3)
    pal = rng.integers(0, 256, size=(4, 3)).astype(np.uint8)
    idx = rng.integers(0, 4, size=(6, 7)).astype(np.uint8)
    raws = bytearray()
    for y in range(6):
        raws.append(0)
        raws.extend(idx[y].tobytes())
    png = (b"\x89PNG\r\n\x1a\n"
           + chunk(b"IHDR", struct.pack(">IIBBBBB", 7, 6, 8, 3, 0, 0, 0))
           + chunk(b"PLTE", pal.tobytes())
           + chunk(b"IDAT", zlib.compress(bytes(raws)))
           + chunk(b"IEND", b""))
    assert (decode_png(png) == pal[idx]).all()


def test_jpeg_codec_roundtrip():
    """Baseline JPEG codec: encode→decode stays within quantization error
    (high PSNR on smooth content, exact on flat blocks); gray + RGB."""
    from mmlspark_amd.io_http.files import decode_image, encode_image
    yy, xx = np.mgrid[0:40, 0:56]
    img = np.dstack([(yy * 2) % 256, (xx * 3) % 256,
                     ((yy + xx) * 2) % 256]).astype(np.uint8)
    dec = decode_image(encode_image(img, "jpg"))
    assert dec.shape == img.shape
    err = (dec.astype(int) - img.astype(int)).astype(float)
    psnr = 10 * np.log10(255 ** 2 / max((err ** 2).mean(), 1e-9))
    assert psnr > 40, psnr
    flat = np.full((16, 24, 3), 201, np.uint8)
    dflat = decode_image(encode_image(flat, "jpeg"))
    assert np.abs(dflat.astype(int) - 201).max() <= 1
    gray = ((yy * 5) % 256).astype(np.uint8)
    dgray = decode_image(encode_image(gray, "jpg"))
    assert dgray.shape == gray.shape
    assert np.abs(dgray.astype(int) - gray.astype(int)).max() <= 4


def test_jpeg_huffman_tables_are_inverse():
    from mmlspark_amd.io_http.jpeg_codec import (HT_AC_C, HT_AC_L, HT_DC_C,
                                                 HT_DC_L, _build_codes,
                                                 _build_decoder)
    for bits, vals in (HT_DC_L, HT_DC_C, HT_AC_L, HT_AC_C):
        enc = _build_codes(bits, vals)
        dec = _build_decoder(bits, vals)
        assert len(enc) == len(vals)
        for v, (code, length) in enc.items():
            assert dec[(length, code)] == v
        # prefix-free: no code is a prefix of a longer one
        codes = sorted(((l, c) for c, l in enc.values()))
        for i, (l1, c1) in enumerate(codes):
            for l2, c2 in codes[i + 1:]:
                if l2 > l1:
                    assert (c2 >> (l2 - l1)) != c1


def test_http_transformer_concurrent_requests():
    """HTTPTransformer with a thread pool (AsyncHTTPClient analog) keeps
    row-to-response alignment under concurrency."""
    from mmlspark_amd.io_http.client import HTTPTransformer
    from mmlspark_amd.io_http.http_schema import HTTPRequestData
    from mmlspark_amd.serving.server import ServingServer

    srv = ServingServer(lambda ps: [{"echo": p["i"]} for p in ps],
                        port=0, mode="continuous").start()
    try:
        n = 24
        df = pd.DataFrame({"req": [
            HTTPRequestData(url=f"http://127.0.0.1:{srv.port}/",
                            method="POST",
                            headers={"Content-Type": "application/json"},
                            entity=json.dumps({"i": i}).encode())
            for i in range(n)]})
        out = HTTPTransformer(inputCol="req", outputCol="resp",
                              concurrency=8).transform(df)
        got = [r.json()["echo"] for r in out["resp"]]
        assert got == list(range(n))  # order preserved despite concurrency
    finally:
        srv.stop()


def test_committed_bindings_are_fresh(tmp_path):
    """The committed R wrappers and Python stubs must match what codegen
    produces from the current registry (staleness guard — the analog of
    the reference's CI assertion that generated wrappers are current)."""
    import filecmp
    import os
    import mmlspark_amd
    mmlspark_amd._register_all()
    from mmlspark_amd.core.codegen import generate_r_wrappers, generate_stubs
    generate_r_wrappers(str(tmp_path / "gen.R"))
    committed = open("bindings/R/mmlspark_amd.R").read()
    assert open(tmp_path / "gen.R").read() == committed, \
        "bindings/R/mmlspark_amd.R is stale — rerun generate_r_wrappers"
    generate_stubs(str(tmp_path / "stubs"))
    for f in sorted(os.listdir(tmp_path / "stubs")):
        assert filecmp.cmp(tmp_path / "stubs" / f,
                           os.path.join("bindings/python_stubs", f),
                           shallow=False), f"stale stub: {f}"
    from mmlspark_amd.core.codegen import generate_docs
    generate_docs(str(tmp_path / "docs"))
    for f in sorted(os.listdir(tmp_path / "docs")):
        assert filecmp.cmp(tmp_path / "docs" / f,
                           os.path.join("docs/api", f),
                           shallow=False), f"stale API doc: {f}"
