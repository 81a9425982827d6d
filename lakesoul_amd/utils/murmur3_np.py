"""Vectorized (numpy) Spark murmur3-32 for fixed-width columns.

Used by the CPU execution path for bucket hashing; the GPU path is the
HIP kernel in csrc/hip/hash.hip. All bit-exact with utils/murmur3.py
(the pure-python oracle) and the reference
``rust/lakesoul-io/src/utils/hash/``.
"""

from __future__ import annotations

import numpy as np

HASH_SEED = 42
_C1 = np.uint32(0xCC9E2D51)
_C2 = np.uint32(0x1B873593)
_M = np.uint32(5)
_N = np.uint32(0xE6546B64)
_F1 = np.uint32(0x85EBCA6B)
_F2 = np.uint32(0xC2B2AE35)


def _rotl(x: np.ndarray, r: int) -> np.ndarray:
    return (x << np.uint32(r)) | (x >> np.uint32(32 - r))


def _mix_k(k: np.ndarray) -> np.ndarray:
    k = k * _C1
    k = _rotl(k, 15)
    k = k * _C2
    return k


def _mix_h(h: np.ndarray, k: np.ndarray) -> np.ndarray:
    h = h ^ k
    h = _rotl(h, 13)
    h = h * _M + _N
    return h


def _finish(h: np.ndarray, nbytes: int) -> np.ndarray:
    h = h ^ np.uint32(nbytes)
    h = h ^ (h >> np.uint32(16))
    h = h * _F1
    h = h ^ (h >> np.uint32(13))
    h = h * _F2
    h = h ^ (h >> np.uint32(16))
    return h


def hash_u32_words(words: np.ndarray, seeds: np.ndarray) -> np.ndarray:
    """Hash rows of 32-bit words: words shape (n, k) uint32; seeds (n,) uint32."""
    with np.errstate(over="ignore"):
        h = seeds.astype(np.uint32).copy()
        for i in range(words.shape[1]):
            h = _mix_h(h, _mix_k(words[:, i].astype(np.uint32)))
        return _finish(h, 4 * words.shape[1])


def hash_column(values: np.ndarray, seeds) -> np.ndarray:
    """Hash one fixed-width column with Spark semantics.

    ``seeds`` may be a scalar (first column) or per-row uint32 array
    (seed chaining for multi-column hashes).
    """
    n = len(values)
    if np.isscalar(seeds):
        seeds = np.full(n, seeds, dtype=np.uint32)
    dt = values.dtype
    with np.errstate(over="ignore"):
        if dt in (np.dtype(np.bool_),):
            w = values.astype(np.uint32).reshape(n, 1)
        elif dt in (np.dtype(np.int8), np.dtype(np.int16), np.dtype(np.int32)):
            # sign-extend to 32-bit (reference: hash mod.rs:53-62)
            w = values.astype(np.int32).view(np.uint32).reshape(n, 1)
        elif dt in (np.dtype(np.uint8), np.dtype(np.uint16), np.dtype(np.uint32)):
            w = values.astype(np.uint32).reshape(n, 1)
        elif dt in (np.dtype(np.int64), np.dtype(np.uint64)):
            v = values.view(np.uint64)
            w = np.empty((n, 2), dtype=np.uint32)
            w[:, 0] = (v & np.uint64(0xFFFFFFFF)).astype(np.uint32)  # low word first (LE)
            w[:, 1] = (v >> np.uint64(32)).astype(np.uint32)
        elif dt == np.dtype(np.float32):
            bits = values.view(np.uint32).copy()
            bits[bits == np.uint32(0x80000000)] = np.uint32(0)  # -0.0 -> 0
            w = bits.reshape(n, 1)
        elif dt == np.dtype(np.float64):
            bits = values.view(np.uint64).copy()
            bits[bits == np.uint64(0x8000000000000000)] = np.uint64(0)
            w = np.empty((n, 2), dtype=np.uint32)
            w[:, 0] = (bits & np.uint64(0xFFFFFFFF)).astype(np.uint32)
            w[:, 1] = (bits >> np.uint64(32)).astype(np.uint32)
        else:
            raise TypeError(f"unsupported numpy dtype for murmur3: {dt}")
        return hash_u32_words(w, seeds)


def create_hashes_np(columns, valid_masks=None) -> np.ndarray:
    """Row hashes over multiple fixed-width columns with seed chaining.

    ``valid_masks``: optional list of boolean arrays (True = non-null);
    NULL leaves the running hash unchanged.
    """
    n = len(columns[0])
    hashes = np.zeros(n, dtype=np.uint32)
    for ci, col in enumerate(columns):
        seeds = hashes if ci >= 1 else np.full(n, HASH_SEED, dtype=np.uint32)
        new = hash_column(np.asarray(col), seeds)
        if valid_masks is not None and valid_masks[ci] is not None:
            mask = np.asarray(valid_masks[ci], dtype=bool)
            hashes = np.where(mask, new, hashes)
        else:
            hashes = new
    return hashes


def bucket_ids_np(hashes: np.ndarray, num_buckets: int) -> np.ndarray:
    return (hashes % np.uint32(num_buckets)).astype(np.uint32)
