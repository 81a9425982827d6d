"""Spark-compatible Murmur3-32 (seed 42) — pure-python reference implementation.

This is the *test oracle*; the production paths are the C++ host
implementation (csrc/cpp/murmur3.h) and the HIP GPU kernel (csrc/hip/hash.hip),
all three bit-exact with the reference's
``rust/lakesoul-io/src/utils/hash/spark_murmur3.rs`` /
``utils/hash/mod.rs`` (and hence with Spark's ``hash()``):

- 32-bit words consumed little-endian;
- tail bytes (len % 4) each processed as a ZERO-extended 32-bit word through
  the full mix (spark_murmur3.rs:56-63);
- finisher XORs the total byte count;
- int8/16/32/bool hash as the sign-extended 32-bit LE word (mod.rs:53-62);
- int64 hashes as 8 LE bytes (mod.rs:64-74);
- float -0.0 hashes as integer 0; otherwise raw IEEE bits (mod.rs:76-90);
- multi-column: column i>=1 re-hashes with the previous hash as seed;
- NULL leaves the running hash unchanged (mod.rs:163-176).
"""

from __future__ import annotations

import struct
from typing import Iterable, Optional, Sequence

MASK = 0xFFFFFFFF
HASH_SEED = 42


def _rotl(x: int, r: int) -> int:
    return ((x << r) | (x >> (32 - r))) & MASK


def _mix_k(k: int) -> int:
    k = (k * 0xCC9E2D51) & MASK
    k = _rotl(k, 15)
    k = (k * 0x1B873593) & MASK
    return k


def _mix_h(h: int, k: int) -> int:
    h ^= k
    h = _rotl(h, 13)
    h = (h * 5 + 0xE6546B64) & MASK
    return h


def hash_bytes(data: bytes, seed: int = HASH_SEED) -> int:
    """Hash a byte string. Tail bytes are zero-extended per byte."""
    h = seed & MASK
    n = len(data)
    nblocks = n // 4
    for i in range(nblocks):
        k = int.from_bytes(data[4 * i : 4 * i + 4], "little")
        h = _mix_h(h, _mix_k(k))
    for b in data[4 * nblocks :]:
        h = _mix_h(h, _mix_k(b))
    h ^= n
    h ^= h >> 16
    h = (h * 0x85EBCA6B) & MASK
    h ^= h >> 13
    h = (h * 0xC2B2AE35) & MASK
    h ^= h >> 16
    return h


def hash_int32(v: int, seed: int = HASH_SEED) -> int:
    """bool/int8/int16/int32 path: sign-extend to 32-bit, 4 LE bytes."""
    return hash_bytes(struct.pack("<I", v & MASK), seed)


def hash_int64(v: int, seed: int = HASH_SEED) -> int:
    return hash_bytes(struct.pack("<Q", v & 0xFFFFFFFFFFFFFFFF), seed)


def hash_float32(v: float, seed: int = HASH_SEED) -> int:
    bits = struct.unpack("<I", struct.pack("<f", v))[0]
    if bits == 0x80000000:  # -0.0 -> 0
        bits = 0
    return hash_bytes(struct.pack("<I", bits), seed)


def hash_float64(v: float, seed: int = HASH_SEED) -> int:
    bits = struct.unpack("<Q", struct.pack("<d", v))[0]
    if bits == 0x8000000000000000:  # -0.0 -> 0
        bits = 0
    return hash_bytes(struct.pack("<Q", bits), seed)


def hash_str(s: str, seed: int = HASH_SEED) -> int:
    return hash_bytes(s.encode("utf-8"), seed)


def hash_value(v, dtype: str, seed: int = HASH_SEED) -> int:
    if dtype in ("bool",):
        return hash_int32(1 if v else 0, seed)
    if dtype in ("int8", "int16", "int32"):
        return hash_int32(int(v), seed)
    if dtype in ("int64",):
        return hash_int64(int(v), seed)
    if dtype == "float32":
        return hash_float32(float(v), seed)
    if dtype == "float64":
        return hash_float64(float(v), seed)
    if dtype in ("string", "utf8", "str"):
        return hash_str(str(v), seed)
    if dtype in ("binary", "bytes"):
        return hash_bytes(bytes(v), seed)
    raise TypeError(f"unsupported dtype for spark murmur3: {dtype}")


def create_hashes(
    columns: Sequence[Iterable],
    dtypes: Sequence[str],
    num_rows: Optional[int] = None,
) -> list:
    """Row hashes over multiple columns with seed chaining
    (reference: utils/hash/mod.rs:304-360). ``None`` cells are NULLs and
    leave the running hash unchanged."""
    cols = [list(c) for c in columns]
    if num_rows is None:
        num_rows = len(cols[0]) if cols else 0
    hashes = [0] * num_rows
    for ci, (col, dt) in enumerate(zip(cols, dtypes)):
        for ri in range(num_rows):
            v = col[ri]
            if v is None:
                continue
            seed = hashes[ri] if ci >= 1 else HASH_SEED
            hashes[ri] = hash_value(v, dt, seed)
    return hashes


def bucket_ids(hashes: Iterable[int], num_buckets: int) -> list:
    """bucket = hash(u32) % num_buckets (reference: reader.rs:188)."""
    return [h % num_buckets for h in hashes]
