"""C ABI tests via ctypes — exercises liblakesoul_amd_c.so exactly the
way a JVM/JNR (or any FFI) consumer would: Arrow C Data Interface
schema/array exchange, merge-on-read reads, writes, murmur3."""

import ctypes
import os

import numpy as np
import pyarrow as pa
import pytest

LIB = os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
                   "lakesoul_amd", "liblakesoul_amd_c.so")


@pytest.fixture(scope="module")
def lib():
    if not os.path.exists(LIB):
        pytest.skip("liblakesoul_amd_c.so not built")
    L = ctypes.CDLL(LIB)
    vp, i64, cp = ctypes.c_void_p, ctypes.c_int64, ctypes.c_char_p
    L.lakesoul_c_reader_create.restype = vp
    L.lakesoul_c_reader_add_file.argtypes = [vp, cp]
    L.lakesoul_c_reader_add_column.argtypes = [vp, cp]
    L.lakesoul_c_reader_add_primary_key.argtypes = [vp, cp]
    L.lakesoul_c_reader_set_batch_size.argtypes = [vp, i64]
    L.lakesoul_c_reader_start.argtypes = [vp]
    L.lakesoul_c_reader_schema.argtypes = [vp, vp]
    L.lakesoul_c_reader_next.argtypes = [vp, vp]
    L.lakesoul_c_reader_close.argtypes = [vp]
    L.lakesoul_c_writer_create.restype = vp
    L.lakesoul_c_writer_create.argtypes = [cp]
    L.lakesoul_c_writer_set_schema.argtypes = [vp, vp]
    L.lakesoul_c_writer_write.argtypes = [vp, vp]
    L.lakesoul_c_writer_close.restype = i64
    L.lakesoul_c_writer_close.argtypes = [vp]
    L.lakesoul_c_last_error.restype = cp
    L.lakesoul_c_murmur3_i64.restype = ctypes.c_uint32
    L.lakesoul_c_murmur3_i64.argtypes = [i64, ctypes.c_uint32]
    L.lakesoul_c_murmur3_bytes.restype = ctypes.c_uint32
    L.lakesoul_c_murmur3_bytes.argtypes = [cp, i64, ctypes.c_uint32]
    return L


def test_murmur3_c_abi(lib):
    from lakesoul_amd.utils import murmur3 as m3

    assert lib.lakesoul_c_murmur3_i64(12345, 42) == m3.hash_int64(12345)
    assert lib.lakesoul_c_murmur3_bytes(b"hello", 5, 42) == m3.hash_bytes(b"hello")


def test_c_writer_then_pyarrow_reads(lib, tmp_path):
    path = str(tmp_path / "cw.parquet").encode()
    w = lib.lakesoul_c_writer_create(path)
    tbl = pa.table(
        {
            "id": pa.array(np.arange(100, dtype=np.int64)),
            "v": pa.array(np.linspace(0, 1, 100)),
            "s": pa.array([f"x{i}" for i in range(100)]),
        }
    )
    # export schema + batch through the Arrow C Data Interface
    c_schema = ctypes.c_void_p()
    c_array = ctypes.c_void_p()

    schema_holder = ctypes.create_string_buffer(72)  # sizeof(ArrowSchema)
    tbl.schema._export_to_c(ctypes.addressof(schema_holder))
    assert lib.lakesoul_c_writer_set_schema(ctypes.c_void_p(w), ctypes.addressof(schema_holder)) == 0, \
        lib.lakesoul_c_last_error()
    batch = tbl.to_batches()[0]
    arr_holder = ctypes.create_string_buffer(80)  # sizeof(ArrowArray)
    struct_arr = batch.to_struct_array()
    struct_arr._export_to_c(ctypes.addressof(arr_holder))
    assert lib.lakesoul_c_writer_write(ctypes.c_void_p(w), ctypes.addressof(arr_holder)) == 0, \
        lib.lakesoul_c_last_error()
    size = lib.lakesoul_c_writer_close(ctypes.c_void_p(w))
    assert size > 0

    import pyarrow.parquet as pq

    got = pq.read_table(path.decode())
    assert got.num_rows == 100
    np.testing.assert_array_equal(got["id"].to_numpy(), np.arange(100))
    assert got["s"].to_pylist() == [f"x{i}" for i in range(100)]


def test_c_reader_merge_on_read(lib, tmp_path, catalog):
    """Write a PK table through the python engine, read it back through
    the C ABI (UseLast merge), import via Arrow C Data."""
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "creader",
        Schema([Field("id", "int64", False), Field("v", "float64", False)]),
        primary_keys=["id"],
        hash_bucket_num=1,
    )
    n = 5000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})
    t.upsert({"id": np.arange(0, n, 4, dtype=np.int64), "v": np.full(len(range(0, n, 4)), 2.5)})
    files = [f.path for f in t.files()]
    assert len(files) == 2

    r = lib.lakesoul_c_reader_create()
    for p in files:
        lib.lakesoul_c_reader_add_file(ctypes.c_void_p(r), p.encode())
    lib.lakesoul_c_reader_add_primary_key(ctypes.c_void_p(r), b"id")
    lib.lakesoul_c_reader_set_batch_size(ctypes.c_void_p(r), 1024)
    assert lib.lakesoul_c_reader_start(ctypes.c_void_p(r)) == 0, lib.lakesoul_c_last_error()

    schema_holder = ctypes.create_string_buffer(72)
    assert lib.lakesoul_c_reader_schema(ctypes.c_void_p(r), ctypes.addressof(schema_holder)) == 0
    schema = pa.Schema._import_from_c(ctypes.addressof(schema_holder))
    names = [f.name for f in schema]
    assert names == ["id", "v"]

    rows = []
    while True:
        arr_holder = ctypes.create_string_buffer(80)
        rc = lib.lakesoul_c_reader_next(ctypes.c_void_p(r), ctypes.addressof(arr_holder))
        assert rc >= 0, lib.lakesoul_c_last_error()
        if rc == 0:
            break
        sa = pa.Array._import_from_c(ctypes.addressof(arr_holder), pa.struct(
            [pa.field("id", pa.int64()), pa.field("v", pa.float64())]
        ))
        rows.append(pa.RecordBatch.from_struct_array(sa))
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    got = pa.Table.from_batches(rows).to_pandas().sort_values("id").reset_index(drop=True)
    assert len(got) == n
    assert (got["v"][::4] == 2.5).all()
    assert (got["v"][1::4] == 0).all()


def test_capi_reader_string_pk(tmp_path, lib):
    """C ABI MOR merge with a string primary key (byte-lexicographic
    ordering, UseLast dedup across files)."""
    import pyarrow.parquet as pq

    f1 = str(tmp_path / "a.parquet")
    f2 = str(tmp_path / "b.parquet")
    # each file writer-sorted by pk, file order = commit order
    pq.write_table(pa.table({
        "k": pa.array(["apple", "kiwi", "pear"], pa.string()),
        "v": pa.array([1.0, 2.0, 3.0], pa.float64()),
    }), f1, use_dictionary=False, compression="zstd")
    pq.write_table(pa.table({
        "k": pa.array(["banana", "kiwi"], pa.string()),
        "v": pa.array([10.0, 20.0], pa.float64()),
    }), f2, use_dictionary=False, compression="zstd")

    r = lib.lakesoul_c_reader_create()
    for f in (f1, f2):
        lib.lakesoul_c_reader_add_file(ctypes.c_void_p(r), f.encode())
    lib.lakesoul_c_reader_add_primary_key(ctypes.c_void_p(r), b"k")
    assert lib.lakesoul_c_reader_start(ctypes.c_void_p(r)) == 0, \
        lib.lakesoul_c_last_error()

    schema_holder = ctypes.create_string_buffer(72)
    assert lib.lakesoul_c_reader_schema(ctypes.c_void_p(r), ctypes.addressof(schema_holder)) == 0
    schema = pa.Schema._import_from_c(ctypes.addressof(schema_holder))
    rows = []
    while True:
        arr_holder = ctypes.create_string_buffer(80)
        rc = lib.lakesoul_c_reader_next(ctypes.c_void_p(r), ctypes.addressof(arr_holder))
        assert rc >= 0, lib.lakesoul_c_last_error()
        if rc == 0:
            break
        sa = pa.Array._import_from_c(ctypes.addressof(arr_holder),
                                     pa.struct(list(schema)))
        rows.append(pa.RecordBatch.from_struct_array(sa))
    lib.lakesoul_c_reader_close(ctypes.c_void_p(r))
    got = pa.Table.from_batches(rows).to_pandas()
    assert got["k"].tolist() == ["apple", "banana", "kiwi", "pear"]
    assert got["v"].tolist() == [1.0, 10.0, 20.0, 3.0]


def test_meta_c_abi_roundtrip(lib, catalog, tmp_path):
    """Metadata C ABI (lakesoul-metadata-c analog): table lookup, snapshot
    file resolution, and an MVCC commit visible to the python client."""
    import json

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "cmeta",
        Schema([Field("id", "int64", False), Field("v", "float64", False)]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    n = 100
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})

    L = ctypes.CDLL(LIB)
    L.lakesoul_meta_open.restype = ctypes.c_void_p
    L.lakesoul_meta_open.argtypes = [ctypes.c_char_p]
    L.lakesoul_meta_table_info.restype = ctypes.c_void_p
    L.lakesoul_meta_table_info.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p]
    L.lakesoul_meta_files_for_latest.restype = ctypes.c_void_p
    L.lakesoul_meta_files_for_latest.argtypes = [ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p]
    L.lakesoul_meta_commit_add_files.restype = ctypes.c_int
    L.lakesoul_meta_commit_add_files.argtypes = [
        ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_int64),
        ctypes.c_int, ctypes.c_char_p]
    L.lakesoul_meta_free_string.argtypes = [ctypes.c_void_p]
    L.lakesoul_meta_close.argtypes = [ctypes.c_void_p]
    L.lakesoul_meta_last_error.restype = ctypes.c_char_p

    h = L.lakesoul_meta_open(t.client.store.path.encode())
    assert h, L.lakesoul_meta_last_error()
    try:
        p = L.lakesoul_meta_table_info(h, b"cmeta", b"default")
        assert p, L.lakesoul_meta_last_error()
        info = json.loads(ctypes.string_at(p).decode())
        L.lakesoul_meta_free_string(p)
        assert info["table_id"] == t.table_id
        assert info["table_schema"]["fields"][0]["name"] == "id"

        p = L.lakesoul_meta_files_for_latest(h, t.table_id.encode(), b"-5")
        files = json.loads(ctypes.string_at(p).decode())
        L.lakesoul_meta_free_string(p)
        py_files = {f.path for f in t.files()}
        assert {f["path"] for f in files} == py_files
        assert all(f["size"] > 0 for f in files)

        # commit an extra file through the C ABI; python must see it
        import pyarrow as pa
        import pyarrow.parquet as pq

        extra = str(tmp_path / "part-cabi000000000_0000.parquet")
        pq.write_table(pa.table({
            "id": pa.array(np.arange(100, 110, dtype=np.int64)),
            "v": pa.array(np.full(10, 5.5)),
        }), extra, use_dictionary=False, compression="zstd")
        paths = (ctypes.c_char_p * 1)(extra.encode())
        sizes = (ctypes.c_int64 * 1)(1234)
        rc = L.lakesoul_meta_commit_add_files(
            h, t.table_id.encode(), b"-5", paths, sizes, 1, b"MergeCommit")
        assert rc == 0, L.lakesoul_meta_last_error()
    finally:
        L.lakesoul_meta_close(h)

    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == n + 10
    assert (df[df.id >= 100]["v"] == 5.5).all()


def test_meta_c_abi_commit_races_python(lib, catalog, tmp_path):
    """C-ABI commits racing python commits: both sides' CAS loops land
    every commit (cross-language MVCC interop)."""
    import threading

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "cmrace",
        Schema([Field("id", "int64", False), Field("v", "float64", False)]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10)})

    L = ctypes.CDLL(LIB)
    L.lakesoul_meta_open.restype = ctypes.c_void_p
    L.lakesoul_meta_open.argtypes = [ctypes.c_char_p]
    L.lakesoul_meta_commit_add_files.restype = ctypes.c_int
    L.lakesoul_meta_commit_add_files.argtypes = [
        ctypes.c_void_p, ctypes.c_char_p, ctypes.c_char_p,
        ctypes.POINTER(ctypes.c_char_p), ctypes.POINTER(ctypes.c_int64),
        ctypes.c_int, ctypes.c_char_p]
    L.lakesoul_meta_close.argtypes = [ctypes.c_void_p]

    import pyarrow as pa
    import pyarrow.parquet as pq

    cfiles = []
    for i in range(4):
        p = str(tmp_path / f"part-crace{i:010d}_0000.parquet")
        pq.write_table(pa.table({
            "id": pa.array(np.arange(100 + i * 10, 110 + i * 10, dtype=np.int64)),
            "v": pa.array(np.full(10, float(i))),
        }), p, use_dictionary=False, compression="zstd")
        cfiles.append(p)

    errs = []

    def c_committer():
        h = L.lakesoul_meta_open(t.client.store.path.encode())
        try:
            for p in cfiles:
                paths = (ctypes.c_char_p * 1)(p.encode())
                sizes = (ctypes.c_int64 * 1)(1)
                if L.lakesoul_meta_commit_add_files(
                        h, t.table_id.encode(), b"-5", paths, sizes, 1,
                        b"MergeCommit") != 0:
                    errs.append("c commit failed")
        finally:
            L.lakesoul_meta_close(h)

    def py_committer():
        try:
            for i in range(4):
                t.upsert({"id": np.array([i], dtype=np.int64),
                          "v": np.array([50.0 + i])})
        except Exception as e:  # pragma: no cover
            errs.append(repr(e))

    th1 = threading.Thread(target=c_committer)
    th2 = threading.Thread(target=py_committer)
    th1.start(); th2.start(); th1.join(); th2.join()
    assert not errs, errs
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == 10 + 40       # 4 C-committed files x 10 rows
    assert (df[df.id == 0]["v"] == 50.0).all()
    assert (df[df.id >= 130]["v"] == 3.0).all()
    # version advanced once per commit: 1 base + 4 py + 4 C
    assert t.latest_version("-5") == 8
