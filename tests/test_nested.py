"""Nested (LIST) columns as first-class table columns:
declare list<T> in a Schema, write real 3-level parquet LISTs, MOR-scan
with UseLast whole-value merge, schema evolution, pyarrow
cross-readability."""

import numpy as np
import pytest
import torch

from lakesoul_amd.io.batch import Batch
from lakesoul_amd.io.schema import Field, Schema, schema_from_json, schema_to_json


def test_list_dtype_canonical():
    f = Field("emb", "list<float32>")
    assert f.dtype == "list<float32>"
    assert not f.is_fixed_width
    f2 = Field("xs", "array<long>")
    assert f2.dtype == "list<int64>"
    f3 = Field("tags", "list<string>")
    assert f3.dtype == "list<string>"
    with pytest.raises(TypeError):
        Field("bad", "list<list<int32>>")


def test_schema_json_roundtrip_with_list():
    s = Schema([Field("id", "int64", False), Field("emb", "list<float32>")])
    j = schema_to_json(s)
    assert "array" in j
    s2 = schema_from_json(j)
    assert s2.field("emb").dtype == "list<float32>"


def test_batch_list_roundtrip_and_take():
    s = Schema([Field("id", "int64", False), Field("v", "list<float32>")])
    lists = [[1.0, 2.0], [], [3.0], None, [4.0, 5.0, 6.0]]
    b = Batch.from_dict({"id": np.arange(5, dtype=np.int64), "v": lists}, s)
    c = b.columns["v"]
    assert c.is_list and len(c) == 5
    assert c.offsets.tolist() == [0, 2, 2, 3, 3, 6]
    assert c.validity.tolist() == [1, 1, 1, 0, 1]
    # arrow round trip
    t = b.to_arrow()
    got = t.column("v").to_pylist()
    assert got[0] == [1.0, 2.0] and got[1] == [] and got[3] is None
    b2 = Batch.from_arrow(t, s)
    assert b2.columns["v"].offsets.tolist() == c.offsets.tolist()
    # take
    sub = b.take(torch.tensor([4, 0, 3]))
    assert sub.columns["v"].offsets.tolist() == [0, 3, 5, 5]
    assert sub.columns["v"].data.tolist() == [4.0, 5.0, 6.0, 1.0, 2.0]
    # slice
    sl = b.slice(1, 4)
    assert sl.columns["v"].offsets.tolist() == [0, 0, 1, 1]


def test_parquet_list_write_read_pyarrow_compat(tmp_path):
    """Our 3-level LIST files read back identically through our reader
    AND through pyarrow (on-disk format compatibility)."""
    import pyarrow.parquet as pq

    from lakesoul_amd.io.writer import _write_batch_to_file_local

    s = Schema([Field("id", "int64", False), Field("emb", "list<float32>")])
    lists = [list(np.arange(i % 4, dtype=np.float32) + i) for i in range(100)]
    lists[7] = None
    lists[13] = []
    b = Batch.from_dict({"id": np.arange(100, dtype=np.int64), "emb": lists}, s)
    path = str(tmp_path / "lists.parquet")
    _write_batch_to_file_local(path, b, "zstd", 1, 50)
    # pyarrow reads the same values
    t = pq.read_table(path)
    got = t.column("emb").to_pylist()
    for i in range(100):
        if lists[i] is None:
            assert got[i] is None, i
        else:
            np.testing.assert_allclose(got[i], lists[i])
    # our own reader (via a temp table scan is covered below); direct file
    from lakesoul_amd.ops import cpp

    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        names = [c["name"] for c in meta["columns"]]
        assert "emb" in names
    finally:
        cpp().close_parquet(h)


def test_table_upsert_mor_scan_with_list(catalog):
    """create/upsert/MOR-scan a table with a list<float32> embedding
    column — the VERDICT #6 done-criterion. UseLast whole-value merge."""
    t = catalog.create_table(
        "emb_table",
        Schema([Field("id", "int64", False), Field("emb", "list<float32>"),
                Field("tag", "string")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    n = 200
    base = [list(np.float32(i) + np.arange(3, dtype=np.float32)) for i in range(n)]
    t.upsert({"id": np.arange(n, dtype=np.int64), "emb": base,
              "tag": [f"t{i}" for i in range(n)]})
    upd_ids = [5, 50, 199]
    t.upsert({"id": np.array(upd_ids, dtype=np.int64),
              "emb": [[9.0, 9.0], None, [7.0]],
              "tag": ["u5", "u50", "u199"]})
    tbl = t.scan().to_arrow()
    df = tbl.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == n
    got = df["emb"].tolist()
    np.testing.assert_allclose(got[5], [9.0, 9.0])
    assert got[50] is None or (isinstance(got[50], float) and np.isnan(got[50]))
    np.testing.assert_allclose(got[199], [7.0])
    np.testing.assert_allclose(got[0], base[0])
    assert df["tag"].iloc[5] == "u5"
    # count + roundtrip through another upsert
    assert t.scan().count() == n


def test_list_schema_evolution_missing_column(catalog):
    """Files written before a list column was added read back as null
    lists (merge nullability rules, reference merge/mod.rs:65-89)."""
    t = catalog.create_table(
        "evolve_l",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10)})
    t.add_columns([Field("emb", "list<float32>")])
    t2 = catalog.table("evolve_l")
    t2.upsert({"id": np.array([3], dtype=np.int64), "v": np.array([1.0]),
               "emb": [[1.0, 2.0]]})
    df = t2.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == 10
    e = df["emb"].tolist()
    np.testing.assert_allclose(e[3], [1.0, 2.0])
    assert e[0] is None or (isinstance(e[0], float) and np.isnan(e[0]))


# ------------------------------------------------------------------ #
# list<string>
# ------------------------------------------------------------------ #

LS_LISTS = [["a", "bb"], [], None, ["ccc"], ["d", "", "ee"]]


def _ls_schema():
    return Schema([Field("id", "int64", False), Field("tags", "list<string>")])


def test_batch_list_string_ops():
    s = _ls_schema()
    b = Batch.from_dict({"id": np.arange(5, dtype=np.int64),
                         "tags": LS_LISTS}, s)
    c = b.columns["tags"]
    assert c.is_list and c.is_list_str and len(c) == 5
    assert c.offsets.tolist() == [0, 2, 2, 2, 3, 6]
    assert c.validity.tolist() == [1, 1, 0, 1, 1]
    # take
    t = b.take(torch.tensor([4, 2, 0]))
    assert t.to_arrow().column("tags").to_pylist() == \
        [["d", "", "ee"], None, ["a", "bb"]]
    # slice
    assert b.slice(1, 4).to_arrow().column("tags").to_pylist() == \
        [[], None, ["ccc"]]
    # concat
    from lakesoul_amd.io.batch import concat_batches

    cc = concat_batches([b, t])
    assert cc.to_arrow().column("tags").to_pylist() == \
        b.to_arrow().column("tags").to_pylist() + \
        t.to_arrow().column("tags").to_pylist()
    # arrow round trip
    b2 = Batch.from_arrow(b.to_arrow(), s)
    assert b2.to_arrow().equals(b.to_arrow())


def test_split_len_prefixed_roundtrip():
    """The C++ parser inverts the PLAIN prefixed stream the CPU scan
    builds for opaque list<string> merging."""
    from lakesoul_amd.ops import cpp

    # rows: ["ab","c"] | [] | ["defg"]
    stream = (b"\x02\x00\x00\x00ab" + b"\x01\x00\x00\x00c" +
              b"" + b"\x04\x00\x00\x00defg")
    row_offs = torch.tensor([0, 11, 11, 19], dtype=torch.int64)
    bys = torch.frombuffer(bytearray(stream), dtype=torch.uint8)
    d = cpp().split_len_prefixed(bys, row_offs)
    assert d["row_offsets"].tolist() == [0, 2, 2, 3]
    assert d["elem_offsets"].tolist() == [0, 2, 3, 7]
    assert bytes(d["bytes"].numpy().tobytes()) == b"abcdefg"


def test_list_string_table_mor(catalog):
    t = catalog.create_table("lstr", _ls_schema(), primary_keys=["id"],
                             hash_bucket_num=2)
    t.upsert({"id": np.arange(5, dtype=np.int64), "tags": LS_LISTS})
    t.upsert({"id": np.array([1, 3], dtype=np.int64),
              "tags": [["x"], ["y", "zz"]]})
    exp = [["a", "bb"], ["x"], None, ["y", "zz"], ["d", "", "ee"]]
    got = t.scan().to_arrow().sort_by("id").column("tags").to_pylist()
    assert got == exp
    # compaction rewrites and re-reads the 3-level BYTE_ARRAY pages
    t.compaction()
    got2 = t.scan().to_arrow().sort_by("id").column("tags").to_pylist()
    assert got2 == exp


def test_list_string_pyarrow_cross_read(catalog):
    """Files we write are standard 3-level LIST of UTF8 — pyarrow reads
    them without us in the loop."""
    import glob

    import pyarrow.parquet as pq

    t = catalog.create_table("lstr_pa", _ls_schema(), primary_keys=["id"],
                             hash_bucket_num=1)
    t.upsert({"id": np.arange(5, dtype=np.int64), "tags": LS_LISTS})
    files = sorted(glob.glob(t.table_path + "/**/*.parquet", recursive=True))
    pt = pq.read_table(files[0]).sort_by("id")
    assert str(pt.schema.field("tags").type).startswith("list")
    assert pt.column("tags").to_pylist() == LS_LISTS


def test_list_string_schema_evolution(catalog):
    t = catalog.create_table(
        "lstr_ev",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1)
    t.upsert({"id": np.arange(4, dtype=np.int64), "v": np.zeros(4)})
    t.add_columns([Field("tags", "list<string>")])
    t2 = catalog.table("lstr_ev")
    t2.upsert({"id": np.array([2], dtype=np.int64), "v": np.array([1.0]),
               "tags": [["new"]]})
    got = t2.scan().to_arrow().sort_by("id").column("tags").to_pylist()
    assert got[2] == ["new"]
    assert got[0] is None and got[3] is None
