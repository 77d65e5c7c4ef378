"""MurmurHash3 x86_32 — VW-compatible feature hashing.

The reference implements the identical hash JVM-side for featurizer/VW parity
(VowpalWabbitMurmurWithPrefix.scala:29-77, hash-equivalence tested against
native murmur in the VW featurizer suites).  Scalar version for strings +
a numpy-vectorized version for bulk uint32 index hashing.
"""
from __future__ import annotations

import numpy as np

_M32 = 0xFFFFFFFF


def _rotl32(x: int, r: int) -> int:
    return ((x << r) | (x >> (32 - r))) & _M32


def murmur3_32(data: bytes, seed: int = 0) -> int:
    """MurmurHash3_x86_32 over bytes (matches VW's uniform_hash)."""
    c1, c2 = 0xCC9E2D51, 0x1B873593
    h = seed & _M32
    n = len(data)
    nblocks = n // 4
    for i in range(nblocks):
        k = int.from_bytes(data[4 * i: 4 * i + 4], "little")
        k = (k * c1) & _M32
        k = _rotl32(k, 15)
        k = (k * c2) & _M32
        h ^= k
        h = _rotl32(h, 13)
        h = (h * 5 + 0xE6546B64) & _M32
    k = 0
    tail = data[nblocks * 4:]
    if len(tail) >= 3:
        k ^= tail[2] << 16
    if len(tail) >= 2:
        k ^= tail[1] << 8
    if len(tail) >= 1:
        k ^= tail[0]
        k = (k * c1) & _M32
        k = _rotl32(k, 15)
        k = (k * c2) & _M32
        h ^= k
    h ^= n
    h ^= h >> 16
    h = (h * 0x85EBCA6B) & _M32
    h ^= h >> 13
    h = (h * 0xC2B2AE35) & _M32
    h ^= h >> 16
    return h


def hash_string(s: str, seed: int = 0) -> int:
    return murmur3_32(s.encode("utf-8"), seed)


def hash_uint32_array(vals: np.ndarray, seed: int = 0) -> np.ndarray:
    """Vectorized murmur3_32 of 4-byte little-endian uint32 values (VW hashes
    pre-hashed integer feature ids this way for hash_all mode)."""
    c1, c2 = np.uint32(0xCC9E2D51), np.uint32(0x1B873593)
    with np.errstate(over="ignore"):
        k = vals.astype(np.uint32) * c1
        k = (k << np.uint32(15)) | (k >> np.uint32(17))
        k = k * c2
        h = np.full(vals.shape, seed, dtype=np.uint32) ^ k
        h = (h << np.uint32(13)) | (h >> np.uint32(19))
        h = h * np.uint32(5) + np.uint32(0xE6546B64)
        h ^= np.uint32(4)
        h ^= h >> np.uint32(16)
        h = h * np.uint32(0x85EBCA6B)
        h ^= h >> np.uint32(13)
        h = h * np.uint32(0xC2B2AE35)
        h ^= h >> np.uint32(16)
    return h
