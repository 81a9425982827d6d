"""Extended C ABI surface (reference lakesoul-io-c lib.rs:113-1324
parity): config option map, merge operators, DSL + Substrait filter
pushdown, CDC rows, async reads, FlushResult — exercised via ctypes AND
via a compiled C consumer (capi_smoke.c) that dlopens the library the
way the reference's JNR Java binding does."""

import ctypes
import os
import subprocess

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "lakesoul_amd", "liblakesoul_amd_c.so")


@pytest.fixture(scope="module")
def lib():
    if not os.path.exists(LIB):
        pytest.skip("liblakesoul_amd_c.so not built")
    L = ctypes.CDLL(LIB)
    vp, i64, cp = ctypes.c_void_p, ctypes.c_int64, ctypes.c_char_p
    L.lakesoul_c_config_create.restype = vp
    L.lakesoul_c_config_add_file.argtypes = [vp, cp]
    L.lakesoul_c_config_add_column.argtypes = [vp, cp]
    L.lakesoul_c_config_add_primary_key.argtypes = [vp, cp]
    L.lakesoul_c_config_add_merge_op.argtypes = [vp, cp, cp]
    L.lakesoul_c_config_add_filter.argtypes = [vp, cp]
    L.lakesoul_c_config_set_filter_substrait.argtypes = [vp, cp, i64]
    L.lakesoul_c_config_set_option.argtypes = [vp, cp, cp]
    L.lakesoul_c_config_get_option.argtypes = [vp, cp]
    L.lakesoul_c_config_get_option.restype = cp
    L.lakesoul_c_config_free.argtypes = [vp]
    L.lakesoul_c_reader_create_from_config.restype = vp
    L.lakesoul_c_reader_create_from_config.argtypes = [vp]
    L.lakesoul_c_reader_start.argtypes = [vp]
    L.lakesoul_c_reader_schema.argtypes = [vp, vp]
    L.lakesoul_c_reader_next.argtypes = [vp, vp]
    L.lakesoul_c_reader_close.argtypes = [vp]
    L.lakesoul_c_last_error.restype = cp
    return L


def _write_sorted_file(path, cols, schema):
    """Write a PK-sorted parquet via the python engine's writer."""
    from lakesoul_amd.io.batch import Batch
    from lakesoul_amd.io.writer import _write_batch_to_file_local

    batch = Batch.from_dict(cols, schema)
    _write_batch_to_file_local(path, batch, "zstd", 1, 250000)


def _read_all(lib, r):
    """Drain the C reader through pyarrow's C-Data import."""
    import pyarrow as pa

    schema_holder = (ctypes.c_byte * 512)()
    assert lib.lakesoul_c_reader_schema(
        ctypes.c_void_p(r), ctypes.addressof(schema_holder)) == 0
    schema = pa.Schema._import_from_c(ctypes.addressof(schema_holder))
    tables = []
    while True:
        arr_holder = (ctypes.c_byte * 512)()
        rc = lib.lakesoul_c_reader_next(ctypes.c_void_p(r), ctypes.addressof(arr_holder))
        assert rc >= 0, lib.lakesoul_c_last_error()
        if rc == 0:
            break
        arr = pa.Array._import_from_c(ctypes.addressof(arr_holder),
                                      pa.struct(list(schema)))
        tables.append(arr)
    import pandas as pd

    if not tables:
        return pd.DataFrame({f.name: [] for f in schema})
    return pd.concat(
        [pa.Table.from_struct_array(a).to_pandas() for a in tables],
        ignore_index=True)


@pytest.fixture
def two_files(tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    schema = Schema([Field("id", "int64", False), Field("v", "float64"),
                     Field("s", "string")])
    f1 = str(tmp_path / "base.parquet")
    f2 = str(tmp_path / "delta.parquet")
    _write_sorted_file(f1, {
        "id": np.arange(100, dtype=np.int64),
        "v": np.full(100, 1.0),
        "s": [f"a{i}" for i in range(100)],
    }, schema)
    _write_sorted_file(f2, {
        "id": np.arange(0, 100, 2, dtype=np.int64),
        "v": np.full(50, 10.0),
        "s": [f"b{i}" for i in range(0, 100, 2)],
    }, schema)
    return f1, f2, schema


def _cfg_reader(lib, files, pks=(), merge_ops=(), filters=(), substrait=None,
                options=()):
    cfg = lib.lakesoul_c_config_create()
    c = ctypes.c_void_p(cfg)
    for f in files:
        lib.lakesoul_c_config_add_file(c, f.encode())
    for p in pks:
        lib.lakesoul_c_config_add_primary_key(c, p.encode())
    for col, op in merge_ops:
        assert lib.lakesoul_c_config_add_merge_op(c, col.encode(), op.encode()) == 0
    for f in filters:
        assert lib.lakesoul_c_config_add_filter(c, f.encode()) == 0, \
            lib.lakesoul_c_last_error()
    if substrait is not None:
        assert lib.lakesoul_c_config_set_filter_substrait(
            c, substrait, len(substrait)) == 0
    for k, v in options:
        lib.lakesoul_c_config_set_option(c, k.encode(), v.encode())
    r = lib.lakesoul_c_reader_create_from_config(c)
    lib.lakesoul_c_config_free(c)
    assert r
    assert lib.lakesoul_c_reader_start(ctypes.c_void_p(r)) == 0, \
        lib.lakesoul_c_last_error()
    return r


def test_merge_operators_c(lib, two_files):
    f1, f2, _ = two_files
    # SumAll on v, UseLastNotNull implicit UseLast on s
    r = _cfg_reader(lib, [f1, f2], pks=["id"],
                    merge_ops=[("v", "SumAll"), ("s", "JoinedAllByComma")])
    df = _read_all(lib, r).sort_values("id").reset_index(drop=True)
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    assert len(df) == 100
    even = df[df["id"] % 2 == 0]
    odd = df[df["id"] % 2 == 1]
    assert (even["v"] == 11.0).all()
    assert (odd["v"] == 1.0).all()
    assert even.iloc[0]["s"] == "a0,b0"
    assert odd.iloc[0]["s"] == "a1"
    # cross-check vs the python CPU oracle
    from lakesoul_amd.io import merge_cpu
    # (semantic parity asserted by values above)


def test_dsl_filter_c(lib, two_files):
    f1, f2, _ = two_files
    r = _cfg_reader(lib, [f1, f2], pks=["id"],
                    filters=["and(gteq(id, 10), lt(id, 20))"])
    df = _read_all(lib, r)
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    assert sorted(df["id"].tolist()) == list(range(10, 20))


def test_string_filter_and_option_map(lib, two_files):
    f1, f2, _ = two_files
    r = _cfg_reader(lib, [f1, f2], pks=["id"],
                    filters=["eq(s, 'b4')"], options=[("batch_size", "16")])
    df = _read_all(lib, r)
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    assert df["id"].tolist() == [4]
    assert df["s"].tolist() == ["b4"]


def test_substrait_filter_c_cross_language(lib, two_files):
    """Substrait bytes produced by the PYTHON encoder decoded by the C++
    engine — the exact path a Spark/Flink connector would use."""
    f1, f2, schema = two_files
    from lakesoul_amd.io.filters import And, Cmp
    from lakesoul_amd.io.substrait import (
        encode_substrait_filter, encode_substrait_plan_filter)

    for enc in (encode_substrait_filter, encode_substrait_plan_filter):
        buf = enc(And(Cmp("id", "gteq", 90), Cmp("v", "gt", 5.0)), schema)
        r = _cfg_reader(lib, [f1, f2], pks=["id"], substrait=buf)
        df = _read_all(lib, r)
        lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
        # v > 5 only on merged even rows (v=10)
        assert sorted(df["id"].tolist()) == [90, 92, 94, 96, 98], enc.__name__


def test_substrait_pyarrow_bytes_into_c(lib, two_files):
    """Substrait bytes from an INDEPENDENT producer (Acero) into the C++
    decoder."""
    pa = pytest.importorskip("pyarrow")
    pc = pytest.importorskip("pyarrow.compute")
    ps = pytest.importorskip("pyarrow.substrait")
    f1, f2, _ = two_files
    schema_pa = pa.schema([("id", pa.int64()), ("v", pa.float64()), ("s", pa.string())])
    buf = bytes(memoryview(ps.serialize_expressions(
        [(pc.field("id") < 6) & (pc.field("v") > 5.0)], ["f"], schema_pa)))
    r = _cfg_reader(lib, [f1, f2], pks=["id"], substrait=buf)
    df = _read_all(lib, r)
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    assert sorted(df["id"].tolist()) == [0, 2, 4]


def test_cdc_rows_dropped_c(lib, tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    schema = Schema([Field("id", "int64", False), Field("v", "float64"),
                     Field("rowKinds", "string")])
    f1 = str(tmp_path / "c1.parquet")
    f2 = str(tmp_path / "c2.parquet")
    _write_sorted_file(f1, {
        "id": np.array([1, 2, 3], dtype=np.int64),
        "v": np.array([1.0, 2.0, 3.0]),
        "rowKinds": ["insert"] * 3,
    }, schema)
    _write_sorted_file(f2, {
        "id": np.array([2], dtype=np.int64),
        "v": np.array([0.0]),
        "rowKinds": ["delete"],
    }, schema)
    r = _cfg_reader(lib, [f1, f2], pks=["id"],
                    options=[("cdc_column", "rowKinds")])
    df = _read_all(lib, r)
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    assert sorted(df["id"].tolist()) == [1, 3]


def test_multi_pk_int_string_c(lib, tmp_path):
    from lakesoul_amd.io.schema import Field, Schema

    schema = Schema([Field("k", "string", False), Field("n", "int64", False),
                     Field("v", "float64")])
    f1 = str(tmp_path / "m1.parquet")
    f2 = str(tmp_path / "m2.parquet")
    _write_sorted_file(f1, {
        "k": ["a", "a", "b"], "n": np.array([1, 2, 1], dtype=np.int64),
        "v": np.array([1.0, 2.0, 3.0]),
    }, schema)
    _write_sorted_file(f2, {
        "k": ["a"], "n": np.array([2], dtype=np.int64),
        "v": np.array([99.0]),
    }, schema)
    r = _cfg_reader(lib, [f1, f2], pks=["k", "n"])
    df = _read_all(lib, r)
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    assert df["v"].tolist() == [1.0, 99.0, 3.0]


def test_compiled_c_consumer(tmp_path):
    """Compile capi_smoke.c with gcc and run it against the library —
    a from-scratch foreign consumer (the JNR-Java stand-in; no JVM in
    this image)."""
    if not os.path.exists(LIB):
        pytest.skip("liblakesoul_amd_c.so not built")
    src = os.path.join(REPO, "csrc", "capi", "tests", "capi_smoke.c")
    exe = str(tmp_path / "capi_smoke")
    subprocess.run(["gcc", "-O2", src, "-o", exe, "-ldl", "-lpthread"],
                   check=True, capture_output=True)
    out = subprocess.run([exe, LIB, str(tmp_path)], capture_output=True,
                         text=True, timeout=120)
    assert out.returncode == 0, out.stdout + out.stderr
    assert "capi_smoke OK" in out.stdout


def test_meta_c_abi_extended_daos(catalog):
    """Extended DAO surface of the metadata C ABI: listing, versioned and
    incremental snapshot queries, lookup by path."""
    import json

    if not os.path.exists(LIB):
        pytest.skip("lib not built")
    L = ctypes.CDLL(LIB)
    vp, cp, i64 = ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int64
    L.lakesoul_meta_open.restype = vp
    L.lakesoul_meta_open.argtypes = [cp]
    for fn in ("lakesoul_meta_list_namespaces",):
        getattr(L, fn).restype = vp
        getattr(L, fn).argtypes = [vp]
    for fn in ("lakesoul_meta_list_tables", "lakesoul_meta_partition_descs",
               "lakesoul_meta_table_info_by_path"):
        getattr(L, fn).restype = vp
        getattr(L, fn).argtypes = [vp, cp]
    L.lakesoul_meta_latest_version.restype = i64
    L.lakesoul_meta_latest_version.argtypes = [vp, cp, cp]
    L.lakesoul_meta_files_for_version.restype = vp
    L.lakesoul_meta_files_for_version.argtypes = [vp, cp, cp, i64]
    L.lakesoul_meta_incremental_files.restype = vp
    L.lakesoul_meta_incremental_files.argtypes = [vp, cp, cp, i64, i64]
    L.lakesoul_meta_free_string.argtypes = [vp]
    L.lakesoul_meta_close.argtypes = [vp]
    L.lakesoul_meta_last_error.restype = cp

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "cdao", Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1)
    t.upsert({"id": np.arange(5, dtype=np.int64), "v": np.zeros(5)})
    t.upsert({"id": np.arange(5, dtype=np.int64), "v": np.ones(5)})

    def grab(p):
        assert p, L.lakesoul_meta_last_error()
        s = ctypes.cast(p, cp).value.decode()
        L.lakesoul_meta_free_string(p)
        return json.loads(s)

    h = L.lakesoul_meta_open(t.client.store.path.encode())
    assert h
    try:
        assert "default" in grab(L.lakesoul_meta_list_namespaces(h))
        assert "cdao" in grab(L.lakesoul_meta_list_tables(h, b"default"))
        descs = grab(L.lakesoul_meta_partition_descs(h, t.table_id.encode()))
        assert descs == ["-5"]
        latest = L.lakesoul_meta_latest_version(h, t.table_id.encode(), b"-5")
        assert latest == 1
        v0 = grab(L.lakesoul_meta_files_for_version(
            h, t.table_id.encode(), b"-5", 0))
        v1 = grab(L.lakesoul_meta_files_for_version(
            h, t.table_id.encode(), b"-5", 1))
        assert len(v0) == 1 and len(v1) == 2
        inc = grab(L.lakesoul_meta_incremental_files(
            h, t.table_id.encode(), b"-5", 0, 1))
        assert len(inc) == 1
        assert inc[0]["path"] == [f for f in v1 if f not in v0][0]["path"]
        info = grab(L.lakesoul_meta_table_info_by_path(
            h, t.table_path.encode()))
        assert info["table_id"] == t.table_id
    finally:
        L.lakesoul_meta_close(h)


def test_meta_c_abi_split_descs(catalog):
    """create_split_desc_array parity (reference lakesoul-metadata-c
    lib.rs:560): one entry per (partition, hash bucket) with ordered
    file paths, PKs and schema — what a JVM connector plans scans from."""
    import json

    if not os.path.exists(LIB):
        pytest.skip("lib not built")
    L = ctypes.CDLL(LIB)
    vp, cp = ctypes.c_void_p, ctypes.c_char_p
    L.lakesoul_meta_open.restype = vp
    L.lakesoul_meta_open.argtypes = [cp]
    L.lakesoul_meta_split_descs.restype = vp
    L.lakesoul_meta_split_descs.argtypes = [vp, cp, cp]
    L.lakesoul_meta_free_string.argtypes = [vp]
    L.lakesoul_meta_close.argtypes = [vp]
    L.lakesoul_meta_last_error.restype = cp

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "csplit", Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=2)
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10)})
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.ones(10)})

    h = L.lakesoul_meta_open(t.client.store.path.encode())
    assert h
    try:
        p = L.lakesoul_meta_split_descs(h, b"csplit", b"default")
        assert p, L.lakesoul_meta_last_error()
        descs = json.loads(ctypes.cast(p, cp).value.decode())
        L.lakesoul_meta_free_string(p)
    finally:
        L.lakesoul_meta_close(h)
    # one split per bucket, both upserts' files in order
    buckets = sorted(d["hash_bucket"] for d in descs)
    assert buckets == [0, 1]
    for d in descs:
        assert d["primary_keys"] == ["id"]
        assert d["partition_desc"] == "-5"
        assert len(d["file_paths"]) == 2
        assert json.loads(d["table_schema"])["type"] == "struct"
        # matches the python scan plan for the same bucket
        units = {u.bucket_id: u.files for u in t.scan().plan()}
        assert [os.path.basename(f) for f in d["file_paths"]] == \
            [os.path.basename(f) for f in units[d["hash_bucket"]]]


def test_meta_c_abi_jwt_interop():
    """C ABI JWT mint/verify interoperates with the gateway's python
    TokenService (same HMAC-SHA256 payload.sig scheme)."""
    import json

    if not os.path.exists(LIB):
        pytest.skip("lib not built")
    L = ctypes.CDLL(LIB)
    vp, cp, i64 = ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int64
    L.lakesoul_meta_jwt_encode.restype = vp
    L.lakesoul_meta_jwt_encode.argtypes = [cp, cp, i64, cp]
    L.lakesoul_meta_jwt_decode.restype = vp
    L.lakesoul_meta_jwt_decode.argtypes = [cp, cp]
    L.lakesoul_meta_free_string.argtypes = [vp]
    L.lakesoul_meta_last_error.restype = cp

    from lakesoul_amd.service.server import TokenService

    ts = TokenService(secret="xsecret")
    # C-minted -> python-verified
    p = L.lakesoul_meta_jwt_encode(b"alice", b"teamA", 3600, b"xsecret")
    assert p, L.lakesoul_meta_last_error()
    tok = ctypes.cast(p, cp).value.decode()
    L.lakesoul_meta_free_string(p)
    claims = ts.verify(tok)
    assert claims["sub"] == "alice" and claims["domain"] == "teamA"
    # python-minted -> C-verified
    tok2 = ts.issue("bob", "teamB")
    p2 = L.lakesoul_meta_jwt_decode(tok2.encode(), b"xsecret")
    assert p2, L.lakesoul_meta_last_error()
    payload = json.loads(ctypes.cast(p2, cp).value.decode())
    L.lakesoul_meta_free_string(p2)
    assert payload["sub"] == "bob" and payload["domain"] == "teamB"
    # tampered signature rejected
    bad = tok[:-2] + ("AA" if not tok.endswith("AA") else "BB")
    assert L.lakesoul_meta_jwt_decode(bad.encode(), b"xsecret") is None
    # wrong secret rejected
    assert L.lakesoul_meta_jwt_decode(tok2.encode(), b"other") is None
