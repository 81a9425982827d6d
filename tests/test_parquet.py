"""Parquet interop: our native writer/reader vs pyarrow (golden oracle).

This is the writer x reader consistency matrix idea from the reference's
``python/tests/compat``, with pyarrow standing in as the
foreign engine.
"""

import os

import numpy as np
import pyarrow as pa
import pyarrow.parquet as pq
import pytest
import torch

from lakesoul_amd.ops import cpp


def _write_ours(path, names, dtypes, columns, offsets=None, validity=None,
                nullable=None, rg=250000, codec=6, level=1):
    n = len(names)
    offsets = offsets or [None] * n
    validity = validity or [None] * n
    if nullable is None:
        nullable = [v is not None for v in validity]
    return cpp().write_parquet(
        str(path), names, dtypes, columns, offsets, validity, nullable, rg, codec, level
    )


def _read_ours_cpu(path):
    """Read a parquet file fully via the native CPU path -> dict of numpy."""
    h = cpp().open_parquet(str(path))
    try:
        meta = cpp().parquet_meta(h)
        cols = {c["name"]: c for c in meta["columns"]}
        out = {}
        for ci, cinfo in enumerate(meta["columns"]):
            parts = []
            masks = []
            str_parts = []
            for rg in range(meta["num_row_groups"]):
                d = cpp().read_chunk_cpu(h, rg, ci)
                nv = d["num_values"]
                dt = cinfo["dtype"]
                if dt in ("string", "binary"):
                    offs = d["offsets"].numpy()
                    bys = d["bytes"].numpy().tobytes()
                    vals = [bys[offs[i]:offs[i + 1]] for i in range(nv)]
                    str_parts.extend(vals)
                else:
                    npdt = {
                        "bool": np.uint8,
                        "int8": np.int32,
                        "int16": np.int32,
                        "int32": np.int32,
                        "int64": np.int64,
                        "float32": np.float32,
                        "float64": np.float64,
                        "date32": np.int32,
                        "timestamp[us]": np.int64,
                        "timestamp[ms]": np.int64,
                    }[dt]
                    parts.append(d["data"].numpy().view(npdt))
                v = d["validity"].numpy()
                masks.append(v if len(v) else np.ones(nv, dtype=np.uint8))
            name = cinfo["name"]
            if cinfo["dtype"] in ("string", "binary"):
                out[name] = (str_parts, np.concatenate(masks))
            else:
                out[name] = (np.concatenate(parts), np.concatenate(masks))
        return out, meta
    finally:
        cpp().close_parquet(h)


def test_roundtrip_ours_to_pyarrow(tmp_path):
    n = 10000
    rng = np.random.default_rng(0)
    ids = np.arange(n, dtype=np.int64)
    vals = rng.normal(size=n)
    f32 = rng.normal(size=n).astype(np.float32)
    i32 = rng.integers(-1000, 1000, n, dtype=np.int32)
    path = tmp_path / "ours.parquet"
    size = _write_ours(
        path,
        ["id", "v", "f", "k"],
        ["int64", "float64", "float32", "int32"],
        [torch.from_numpy(ids), torch.from_numpy(vals), torch.from_numpy(f32), torch.from_numpy(i32)],
        nullable=[False, False, False, False],
    )
    assert size > 0
    t = pq.read_table(str(path))
    assert t.num_rows == n
    np.testing.assert_array_equal(t["id"].to_numpy(), ids)
    np.testing.assert_array_equal(t["v"].to_numpy(), vals)
    np.testing.assert_array_equal(t["f"].to_numpy(), f32)
    np.testing.assert_array_equal(t["k"].to_numpy(), i32)


def test_roundtrip_with_nulls_and_strings(tmp_path):
    n = 5000
    rng = np.random.default_rng(1)
    ids = np.arange(n, dtype=np.int64)
    vals = rng.normal(size=n)
    mask = (rng.random(n) > 0.3).astype(np.uint8)
    strings = [f"row-{i}" if i % 3 else "" for i in range(n)]
    joined = "".join(strings).encode()
    offs = np.zeros(n + 1, dtype=np.int32)
    offs[1:] = np.cumsum([len(s.encode()) for s in strings]).astype(np.int32)
    path = tmp_path / "ours2.parquet"
    _write_ours(
        path,
        ["id", "v", "s"],
        ["int64", "float64", "string"],
        [
            torch.from_numpy(ids),
            torch.from_numpy(vals),
            torch.from_numpy(np.frombuffer(joined, dtype=np.uint8).copy()),
        ],
        offsets=[None, None, torch.from_numpy(offs)],
        validity=[None, torch.from_numpy(mask), None],
        nullable=[False, True, False],
    )
    t = pq.read_table(str(path))
    assert t.num_rows == n
    np.testing.assert_array_equal(t["id"].to_numpy(), ids)
    got_v = t["v"].to_pylist()
    for i in range(n):
        if mask[i]:
            assert got_v[i] == pytest.approx(vals[i])
        else:
            assert got_v[i] is None
    assert t["s"].to_pylist() == strings


def test_roundtrip_bool_and_small_ints(tmp_path):
    n = 1000
    rng = np.random.default_rng(2)
    flags = (rng.random(n) > 0.5).astype(np.uint8)
    i16 = rng.integers(-300, 300, n, dtype=np.int16).astype(np.int32)
    path = tmp_path / "ours3.parquet"
    _write_ours(
        path,
        ["flag", "small"],
        ["bool", "int16"],
        [torch.from_numpy(flags), torch.from_numpy(i16)],
        nullable=[False, False],
    )
    t = pq.read_table(str(path))
    np.testing.assert_array_equal(t["flag"].to_numpy(), flags.astype(bool))
    assert t["small"].type == pa.int16()
    np.testing.assert_array_equal(t["small"].to_numpy().astype(np.int32), i16)


@pytest.mark.parametrize("compression", ["zstd", "snappy", "none"])
@pytest.mark.parametrize("dictionary", [True, False])
def test_read_pyarrow_files(tmp_path, compression, dictionary):
    n = 20000
    rng = np.random.default_rng(3)
    tbl = pa.table(
        {
            "id": pa.array(np.arange(n, dtype=np.int64)),
            "v": pa.array(rng.normal(size=n)),
            "cat": pa.array((np.arange(n) % 50).astype(np.int32)),
            "s": pa.array([f"name_{i % 100}" for i in range(n)]),
        }
    )
    path = tmp_path / "pa.parquet"
    pq.write_table(
        tbl,
        str(path),
        compression=compression,
        use_dictionary=dictionary,
        row_group_size=7000,
    )
    data, meta = _read_ours_cpu(path)
    assert meta["num_rows"] == n
    np.testing.assert_array_equal(data["id"][0], np.arange(n, dtype=np.int64))
    np.testing.assert_allclose(data["v"][0], tbl["v"].to_numpy())
    np.testing.assert_array_equal(data["cat"][0], tbl["cat"].to_numpy())
    assert [b.decode() for b in data["s"][0]] == tbl["s"].to_pylist()


def test_read_pyarrow_with_nulls(tmp_path):
    n = 9999
    rng = np.random.default_rng(4)
    raw = rng.normal(size=n)
    vals = [None if rng.random() < 0.25 else float(raw[i]) for i in range(n)]
    tbl = pa.table({"id": np.arange(n, dtype=np.int64), "v": pa.array(vals, type=pa.float64())})
    path = tmp_path / "pa_nulls.parquet"
    pq.write_table(tbl, str(path), compression="zstd", use_dictionary=False, row_group_size=4000)
    data, meta = _read_ours_cpu(path)
    got, mask = data["v"]
    for i in range(n):
        if vals[i] is None:
            assert mask[i] == 0
        else:
            assert mask[i] == 1
            assert got[i] == pytest.approx(vals[i])


def test_multiple_row_groups_and_stats(tmp_path):
    n = 600000  # > 2 row groups at 250k
    ids = np.arange(n, dtype=np.int64)
    path = tmp_path / "rg.parquet"
    _write_ours(path, ["id"], ["int64"], [torch.from_numpy(ids)], nullable=[False])
    h = cpp().open_parquet(str(path))
    meta = cpp().parquet_meta(h)
    assert meta["num_row_groups"] == 3
    s0 = meta["row_groups"][0]["columns"][0]
    assert np.frombuffer(s0["min"], dtype=np.int64)[0] == 0
    assert np.frombuffer(s0["max"], dtype=np.int64)[0] == 249999
    cpp().close_parquet(h)
    # pyarrow agrees
    pf = pq.ParquetFile(str(path))
    assert pf.metadata.num_row_groups == 3
    assert pf.metadata.row_group(0).column(0).statistics.min == 0

def test_decimal_roundtrip_ours_to_pyarrow(tmp_path):
    """decimal(p,s) written as INT64 unscaled with DECIMAL logical type;
    pyarrow must read it back as decimal128 with the right values."""
    import decimal

    unscaled = np.array([12345, -995, 0, 10**10], dtype=np.int64)
    path = tmp_path / "dec.parquet"
    _write_ours(path, ["id", "amt"], ["int64", "decimal(12,2)"],
                [torch.arange(4, dtype=torch.int64), torch.from_numpy(unscaled)])
    t = pq.read_table(path)
    assert t.schema.field("amt").type == pa.decimal128(12, 2)
    assert t.column("amt").to_pylist() == [
        decimal.Decimal(int(x)).scaleb(-2) for x in unscaled
    ]


def test_decimal_roundtrip_pyarrow_to_ours(tmp_path):
    """pyarrow writing decimal as integer (store_decimal_as_integer) must be
    readable by our reader with matching unscaled values + validity."""
    import decimal

    vals = [decimal.Decimal("123.45"), decimal.Decimal("-9.99"), None,
            decimal.Decimal("0.01")]
    t = pa.table({"id": pa.array([1, 2, 3, 4], pa.int64()),
                  "amt": pa.array(vals, pa.decimal128(10, 2))})
    path = tmp_path / "pad.parquet"
    pq.write_table(t, path, store_decimal_as_integer=True,
                   use_dictionary=False, compression="zstd")

    h = cpp().open_parquet(str(path))
    try:
        meta = cpp().parquet_meta(h)
        amt_ci = [i for i, c in enumerate(meta["columns"]) if c["name"] == "amt"][0]
        assert meta["columns"][amt_ci]["dtype"] == "decimal(10,2)"
        d = cpp().read_chunk_cpu(h, 0, amt_ci)
        got = d["data"].numpy().view(np.int64)
        mask = d["validity"].numpy()
    finally:
        cpp().close_parquet(h)
    assert mask.tolist() == [1, 1, 0, 1]
    np.testing.assert_array_equal(got[[0, 1, 3]], [12345, -999, 1])


def test_decimal_batch_arrow_bridge():
    """Batch.from_arrow / to_arrow map decimal128 <-> unscaled int64."""
    import decimal

    from lakesoul_amd.io.batch import Batch
    from lakesoul_amd.io.schema import Field, Schema

    vals = [decimal.Decimal("1.50"), None, decimal.Decimal("-2.25")]
    t = pa.table({"amt": pa.array(vals, pa.decimal128(8, 2))})
    sch = Schema([Field("amt", "decimal(8,2)")])
    b = Batch.from_arrow(t, sch)
    np.testing.assert_array_equal(b.columns["amt"].data.numpy()[[0, 2]], [150, -225])
    assert b.columns["amt"].validity.numpy().tolist() == [1, 0, 1]
    assert b.to_arrow().column("amt").to_pylist() == vals


def test_flba_read(tmp_path):
    """FLBA (fixed-length byte array) columns from foreign writers read
    as binary cells."""
    vals = [b"0123456789abcdef", b"ffffffffffffffff", b"0000000000000000"]
    t = pa.table({"id": pa.array([1, 2, 3], pa.int64()),
                  "u": pa.array(vals, pa.binary(16))})
    path = str(tmp_path / "flba.parquet")
    pq.write_table(t, path, use_dictionary=False, compression="zstd")
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        ci = [i for i, c in enumerate(meta["columns"]) if c["name"] == "u"][0]
        assert meta["columns"][ci]["dtype"] == "binary"
        d = cpp().read_chunk_cpu(h, 0, ci)
        offs = d["offsets"].numpy()
        bys = d["bytes"].numpy().tobytes()
        got = [bys[offs[i]:offs[i + 1]] for i in range(3)]
    finally:
        cpp().close_parquet(h)
    assert got == vals


def test_flba_read_dict_encoded(tmp_path):
    vals = [b"aaaaaaaa", b"bbbbbbbb"] * 50
    t = pa.table({"u": pa.array(vals, pa.binary(8))})
    path = str(tmp_path / "flbad.parquet")
    pq.write_table(t, path, use_dictionary=True, compression="snappy")
    h = cpp().open_parquet(path)
    try:
        d = cpp().read_chunk_cpu(h, 0, 0)
        offs = d["offsets"].numpy()
        bys = d["bytes"].numpy().tobytes()
        got = [bys[offs[i]:offs[i + 1]] for i in range(100)]
    finally:
        cpp().close_parquet(h)
    assert got == vals


def test_int96_read(tmp_path):
    """Legacy INT96 timestamps (spark/impala) read as timestamp[ns]."""
    import datetime

    ts = [datetime.datetime(2020, 1, 1, 12, 0, 0),
          datetime.datetime(1999, 12, 31, 23, 59, 59),
          datetime.datetime(2026, 9, 11, 1, 2, 3, 456789)]
    t = pa.table({"t": pa.array(ts, pa.timestamp("us"))})
    path = str(tmp_path / "i96.parquet")
    pq.write_table(t, path, use_deprecated_int96_timestamps=True,
                   use_dictionary=False, compression="zstd")
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        assert meta["columns"][0]["dtype"] == "timestamp[ns]"
        d = cpp().read_chunk_cpu(h, 0, 0)
        got = d["data"].numpy().view(np.int64)
    finally:
        cpp().close_parquet(h)
    expect = np.array([int(x.replace(tzinfo=datetime.timezone.utc).timestamp() * 1e6) * 1000
                       for x in ts], dtype=np.int64)
    np.testing.assert_array_equal(got, expect)


def _read_list_col(path, name):
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        ci = [i for i, c in enumerate(meta["columns"]) if c["name"] == name][0]
        dt = meta["columns"][ci]["dtype"]
        parts = []
        for rg in range(meta["num_row_groups"]):
            parts.append(cpp().read_chunk_cpu(h, rg, ci))
        return dt, parts
    finally:
        cpp().close_parquet(h)


def test_list_int64_read(tmp_path):
    """Standard 3-level LIST<int64>: rep/def level decode vs pyarrow."""
    data = [[1, 2, 3], [], None, [4], [5, None, 7], [8, 9]]
    t = pa.table({"l": pa.array(data, pa.list_(pa.int64()))})
    path = str(tmp_path / "list.parquet")
    pq.write_table(t, path, use_dictionary=False, compression="zstd")
    dt, parts = _read_list_col(path, "l")
    assert dt == "list<int64>"
    d = parts[0]
    offs = d["list_offsets"].numpy()
    lv = d["list_validity"].numpy()
    vals = d["data"].numpy().view(np.int64)
    ev = d["validity"].numpy()
    got = []
    for r in range(len(offs) - 1):
        if not lv[r]:
            got.append(None)
            continue
        row = []
        for e in range(offs[r], offs[r + 1]):
            row.append(None if (len(ev) and not ev[e]) else int(vals[e]))
        got.append(row)
    assert got == data


def test_list_string_dict_read(tmp_path):
    data = [["a", "bb"], ["a"], [], ["ccc", "a", "bb"]]
    t = pa.table({"s": pa.array(data, pa.list_(pa.string()))})
    path = str(tmp_path / "lists.parquet")
    pq.write_table(t, path, use_dictionary=True, compression="snappy")
    dt, parts = _read_list_col(path, "s")
    assert dt == "list<string>"
    d = parts[0]
    offs = d["list_offsets"].numpy()
    soffs = d["offsets"].numpy()
    bys = d["bytes"].numpy().tobytes()
    got = []
    for r in range(len(offs) - 1):
        row = [bys[soffs[e]:soffs[e + 1]].decode() for e in range(offs[r], offs[r + 1])]
        got.append(row)
    assert got == data


def test_list_multirowgroup_and_longrows(tmp_path):
    rng = np.random.default_rng(0)
    data = [list(map(int, rng.integers(0, 100, int(rng.integers(0, 30)))))
            for _ in range(5000)]
    t = pa.table({"l": pa.array(data, pa.list_(pa.int64()))})
    path = str(tmp_path / "listbig.parquet")
    pq.write_table(t, path, use_dictionary=False, compression="zstd",
                   row_group_size=1700)
    dt, parts = _read_list_col(path, "l")
    got = []
    for d in parts:
        offs = d["list_offsets"].numpy()
        vals = d["data"].numpy().view(np.int64)
        for r in range(len(offs) - 1):
            got.append([int(vals[e]) for e in range(offs[r], offs[r + 1])])
    assert got == data


def test_struct_read_flattened(tmp_path):
    """Plain struct columns flatten to dotted leaf columns with correct
    multi-level def decoding (null struct vs null field)."""
    t = pa.table({
        "id": pa.array([1, 2, 3, 4], pa.int64()),
        "s": pa.array(
            [{"a": 10, "b": "x"}, None, {"a": None, "b": "y"}, {"a": 40, "b": None}],
            pa.struct([("a", pa.int64()), ("b", pa.string())])),
    })
    path = str(tmp_path / "struct.parquet")
    pq.write_table(t, path, use_dictionary=False, compression="zstd")
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        names = [c["name"] for c in meta["columns"]]
        assert names == ["id", "s.a", "s.b"]
        ai = names.index("s.a")
        bi = names.index("s.b")
        da = cpp().read_chunk_cpu(h, 0, ai)
        db = cpp().read_chunk_cpu(h, 0, bi)
    finally:
        cpp().close_parquet(h)
    av = da["data"].numpy().view(np.int64)
    avm = da["validity"].numpy()
    assert avm.tolist() == [1, 0, 0, 1]
    assert av[0] == 10 and av[3] == 40
    bvm = db["validity"].numpy()
    assert bvm.tolist() == [1, 0, 1, 0]
    offs = db["offsets"].numpy()
    bys = db["bytes"].numpy().tobytes()
    vals = [bys[offs[i]:offs[i + 1]].decode() for i in range(4)]
    assert vals[0] == "x" and vals[2] == "y"


def test_map_read_as_parallel_lists(tmp_path):
    """MAP columns surface as two parallel list columns (m.key/m.value)
    with shared row ranges."""
    data = [{"a": 1, "b": 2}, {}, None, {"c": None, "d": 4}]
    t = pa.table({"m": pa.array(data, pa.map_(pa.string(), pa.int64()))})
    path = str(tmp_path / "map.parquet")
    pq.write_table(t, path, use_dictionary=False, compression="zstd")
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        names = [c["name"] for c in meta["columns"]]
        assert names == ["m.key", "m.value"]
        assert meta["columns"][0]["dtype"] == "list<string>"
        assert meta["columns"][1]["dtype"] == "list<int64>"
        dk = cpp().read_chunk_cpu(h, 0, 0)
        dv = cpp().read_chunk_cpu(h, 0, 1)
    finally:
        cpp().close_parquet(h)
    ko, vo = dk["list_offsets"].numpy(), dv["list_offsets"].numpy()
    np.testing.assert_array_equal(ko, vo)
    kv = dk["list_validity"].numpy()
    assert kv.tolist() == [1, 1, 0, 1]
    soffs = dk["offsets"].numpy()
    bys = dk["bytes"].numpy().tobytes()
    keys = [bys[soffs[i]:soffs[i + 1]].decode() for i in range(4)]
    assert keys == ["a", "b", "c", "d"]
    vals = dv["data"].numpy().view(np.int64)
    vvm = dv["validity"].numpy()
    assert vvm.tolist() == [1, 1, 0, 1]
    assert vals[0] == 1 and vals[3] == 4


def test_datapage_v2_write_roundtrip(tmp_path, monkeypatch):
    """LAKESOUL_PAGE_V2=1 emits DataPageV2; pyarrow and our reader agree.
    NOTE: the v2 flag is read once per process (static init) — this test
    spawns a subprocess with the env set."""
    import subprocess
    import sys
    import textwrap

    script = textwrap.dedent(f"""
        import sys, numpy as np, torch
        sys.path.insert(0, {str(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))!r})
        from lakesoul_amd.ops import cpp
        n = 10000
        rng = np.random.default_rng(0)
        v = rng.normal(size=n)
        validity = (rng.random(n) > 0.1).astype(np.uint8)
        cpp().write_parquet(
            {str(tmp_path / 'v2.parquet')!r}, ["a", "b"], ["int64", "float64"],
            [torch.arange(n, dtype=torch.int64), torch.from_numpy(v)],
            [None, None], [None, torch.from_numpy(validity)], [False, True],
            4000, 6, 1)
        print("WROTE")
    """)
    env = dict(os.environ, LAKESOUL_PAGE_V2="1")
    r = subprocess.run([sys.executable, "-c", script], capture_output=True,
                       text=True, env=env, timeout=300)
    assert "WROTE" in r.stdout, r.stderr[-800:]

    t = pq.read_table(str(tmp_path / "v2.parquet"))
    assert t.num_rows == 10000
    np.testing.assert_array_equal(t.column("a").to_numpy(), np.arange(10000))
    # our reader handles the v2 pages too
    h = cpp().open_parquet(str(tmp_path / "v2.parquet"))
    try:
        got_parts = []
        meta = cpp().parquet_meta(h)
        for rg in range(meta["num_row_groups"]):
            d = cpp().read_chunk_cpu(h, rg, 0)
            got_parts.append(d["data"].numpy().view(np.int64))
        np.testing.assert_array_equal(np.concatenate(got_parts), np.arange(10000))
        nulls = 0
        for rg in range(meta["num_row_groups"]):
            d = cpp().read_chunk_cpu(h, rg, 1)
            vmask = d["validity"].numpy()
            nulls += int((vmask == 0).sum()) if len(vmask) else 0
        assert nulls == int((t.column("b").null_count))
    finally:
        cpp().close_parquet(h)
