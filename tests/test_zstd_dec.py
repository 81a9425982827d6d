"""Differential tests: the from-scratch zstd decoder (csrc/cpp/zstd_dec.h)
vs libzstd-compressed data — random, structured (parquet-page-like), text,
multiple levels and sizes. libzstd (via our writer) is the golden encoder."""

import numpy as np
import pytest

from lakesoul_amd.ops import cpp


def _compress(data: bytes, level: int = 1) -> bytes:
    """Compress with libzstd through the C++ extension's writer codec
    (the exact encoder whose output the GPU decoder must handle)."""
    return cpp().zstd_compress_ref(data, level)


def _check(data: bytes, level=1):
    comp = _compress(data, level)
    out = cpp().zstd_decode_ref(comp, max(1, len(data)))
    assert out == data, (
        f"mismatch: n={len(data)} level={level} "
        f"first_diff={next((i for i in range(min(len(out), len(data))) if out[i] != data[i]), None)} "
        f"lens {len(out)} vs {len(data)}"
    )


def test_empty_and_tiny():
    for n in (0, 1, 2, 3, 7, 16, 63):
        _check(bytes(range(n)))


def test_rle_like():
    _check(b"\x00" * 100000)
    _check(b"ab" * 50000)
    _check(b"x" * 131072 * 3)  # multi-block


def test_random_bytes_incompressible():
    rng = np.random.default_rng(0)
    for n in (100, 4096, 70000, 300000):
        _check(rng.integers(0, 256, n, dtype=np.uint8).tobytes())


def test_structured_int64_pages():
    """What parquet PLAIN int64 column pages actually look like."""
    rng = np.random.default_rng(1)
    for level in (1, 3):
        ids = np.arange(100000, dtype=np.int64)
        _check(ids.tobytes(), level)
        vals = rng.normal(size=65536)
        _check(vals.tobytes(), level)
        small = rng.integers(0, 1000, 80000).astype(np.int32)
        _check(small.tobytes(), level)


def test_text_like():
    words = ["lake", "soul", "gpu", "parquet", "zstd", "merge", "bucket"]
    rng = np.random.default_rng(2)
    txt = " ".join(words[i] for i in rng.integers(0, len(words), 200000))
    for level in (1, 2, 3):
        _check(txt.encode(), level)


def test_real_writer_column_bytes(tmp_path):
    """The exact byte patterns our parquet writer compresses per page:
    PLAIN-encoded column values of the bench schema."""
    import torch

    rng = np.random.default_rng(3)
    n = 250000
    path = str(tmp_path / "z.parquet")
    cpp().write_parquet(
        path, ["a", "b", "c"], ["int64", "float64", "int32"],
        [torch.from_numpy(np.arange(n, dtype=np.int64)),
         torch.from_numpy(rng.normal(size=n)),
         torch.from_numpy(rng.integers(0, 500, n).astype(np.int32))],
        [None] * 3, [None] * 3, [False] * 3, 100000, 6, 1,
    )
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        for rg in range(meta["num_row_groups"]):
            for ci in range(len(meta["columns"])):
                d = cpp().read_chunk_raw(h, rg, ci)
                raw = bytes(d["values"].numpy().tobytes())
                _check(raw, 1)
    finally:
        cpp().close_parquet(h)


def test_fuzz_slices():
    """Many small windows of mixed content exercise rare paths (RLE
    literals, treeless repeats, single-stream huffman, raw literals)."""
    rng = np.random.default_rng(4)
    base = bytearray()
    base += bytes(rng.integers(0, 256, 5000, dtype=np.uint8))
    base += b"A" * 3000
    base += np.arange(2000, dtype=np.int16).tobytes()
    base += bytes(rng.integers(65, 70, 8000, dtype=np.uint8))
    for trial in range(200):
        a = int(rng.integers(0, len(base) - 1))
        b = int(rng.integers(a + 1, len(base) + 1))
        lvl = int(rng.integers(1, 6))
        _check(bytes(base[a:b]), lvl)


def test_large_multiblock_high_level():
    """1 MB+ payloads at high levels: multi-block frames with matches
    reaching back into earlier blocks, long offsets, big windows."""
    rng = np.random.default_rng(5)
    words = [bytes(rng.integers(0, 256, int(rng.integers(4, 50)), dtype=np.uint8))
             for _ in range(500)]
    blob = b"".join(words[i] for i in rng.integers(0, 500, 60000))
    for level in (1, 9, 19):
        _check(blob[: 1 << 20], level)
    # sorted integers (delta-friendly, long matches)
    _check(np.arange(1 << 17, dtype=np.int64).tobytes(), 19)


def test_fuzz_random_sizes_levels():
    rng = np.random.default_rng(6)
    for trial in range(60):
        n = int(rng.integers(1, 200000))
        kind = trial % 3
        if kind == 0:
            data = bytes(rng.integers(0, 256, n, dtype=np.uint8))
        elif kind == 1:
            data = bytes(rng.integers(0, 4, n, dtype=np.uint8))
        else:
            data = (np.cumsum(rng.integers(0, 3, n // 8 + 1))
                    .astype(np.int64).tobytes()[:n])
        _check(data, int(rng.integers(1, 20)))
