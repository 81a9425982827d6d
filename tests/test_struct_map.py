"""struct<...> and map<K,V> as declarable table columns.

The reference supports arbitrary nested Arrow schemas through arrow-rs;
here structs store as parquet groups of scalar leaves (optional group +
required members, so each leaf encodes exactly like a flat nullable
column) and maps as the standard MAP shape (group{MAP}/key_value with
list-element levels). Scans read the dotted leaves through the ordinary
MOR merge and reassemble the top-level column
(reference: rust/lakesoul-io schema handling; parquet.thrift nested
encodings)."""

import numpy as np
import pytest
import torch

from lakesoul_amd.io.batch import Batch, concat_batches
from lakesoul_amd.io.schema import (Field, Schema, map_params,
                                    schema_from_json, schema_to_json,
                                    struct_members)

ST = "struct<a:int64,b:string>"
MP = "map<string,int64>"

ROWS_ST = [{"a": 1, "b": "x"}, None, {"a": 3, "b": "zz"}, {"a": 4, "b": ""}]
ROWS_M = [{"k1": 1, "k2": 2}, None, {}, {"only": 7}]


def _schema():
    return Schema([Field("id", "int64", False), Field("st", ST),
                   Field("m", MP)])


def _mk_batch(n=4):
    return Batch.from_dict({"id": np.arange(n, dtype=np.int64),
                            "st": ROWS_ST[:n], "m": ROWS_M[:n]}, _schema())


def _pairs(rows):
    return [None if r is None else list(r.items()) for r in rows]


def test_dtype_parsing_and_serde():
    assert struct_members("struct<a:long, b:decimal(10,2)>") == [
        ("a", "long"), ("b", "decimal(10,2)")]
    assert map_params("map<string, int>") == ("string", "int")
    from lakesoul_amd.io.schema import canonical_dtype

    assert canonical_dtype("struct<a:long,b:decimal(10,2)>") == \
        "struct<a:int64,b:decimal(10,2)>"
    assert canonical_dtype("map<int,str>") == "map<int32,string>"
    with pytest.raises(TypeError):
        canonical_dtype("struct<a:list<int32>>")
    with pytest.raises(TypeError):
        canonical_dtype("map<string,decimal(5,2)>")
    # spark JSON round trip (nested struct/map type objects)
    s = _schema()
    s2 = schema_from_json(schema_to_json(s))
    assert [f.dtype for f in s2] == [f.dtype for f in s]


def test_batch_ops_struct_map():
    b = _mk_batch()
    t = b.to_arrow()
    assert t.column("st").to_pylist() == ROWS_ST
    assert t.column("m").to_pylist() == _pairs(ROWS_M)
    # arrow round trip
    b2 = Batch.from_arrow(t, _schema())
    assert b2.to_arrow().equals(t)
    # take
    tk = b.take(torch.tensor([3, 1, 0]))
    assert tk.to_arrow().column("st").to_pylist() == \
        [ROWS_ST[3], None, ROWS_ST[0]]
    assert tk.to_arrow().column("m").to_pylist() == \
        [_pairs(ROWS_M)[3], None, _pairs(ROWS_M)[0]]
    # slice
    sl = b.slice(1, 3)
    assert sl.to_arrow().column("st").to_pylist() == ROWS_ST[1:3]
    # concat
    cc = concat_batches([b, tk])
    assert cc.to_arrow().column("st").to_pylist() == \
        ROWS_ST + [ROWS_ST[3], None, ROWS_ST[0]]
    assert cc.num_rows == 7


def test_parquet_write_pyarrow_cross_read(tmp_path):
    import pyarrow.parquet as pq

    from lakesoul_amd.io.writer import _write_batch_to_file_local

    b = _mk_batch()
    path = str(tmp_path / "sm.parquet")
    _write_batch_to_file_local(path, b, "zstd", 1, 2)
    t = pq.read_table(path)
    assert t.column("st").to_pylist() == ROWS_ST
    assert t.column("m").to_pylist() == _pairs(ROWS_M)
    assert str(t.schema.field("st").type).startswith("struct")
    assert str(t.schema.field("m").type).startswith("map")


def test_table_mor_struct_map(catalog):
    t = catalog.create_table("sm", _schema(), primary_keys=["id"],
                             hash_bucket_num=2)
    n = 20
    st0 = [{"a": i, "b": f"s{i}"} if i % 5 else None for i in range(n)]
    m0 = [None if i % 7 == 0 else
          {f"k{j}": i * 10 + j for j in range(i % 3)} for i in range(n)]
    t.upsert({"id": np.arange(n, dtype=np.int64), "st": st0, "m": m0})
    t.upsert({"id": np.array([1, 3], dtype=np.int64),
              "st": [{"a": 100, "b": "upd"}, None],
              "m": [{"z": 9}, {"y": 8}]})
    exp_st = list(st0)
    exp_st[1] = {"a": 100, "b": "upd"}
    exp_st[3] = None
    exp_m = _pairs(m0)
    exp_m[1] = [("z", 9)]
    exp_m[3] = [("y", 8)]
    at = t.scan().to_arrow().sort_by("id")
    assert at.column("st").to_pylist() == exp_st
    assert at.column("m").to_pylist() == exp_m
    t.compaction()
    at2 = t.scan().to_arrow().sort_by("id")
    assert at2.column("st").to_pylist() == exp_st
    assert at2.column("m").to_pylist() == exp_m
    # projection keeps reassembly working with a subset of leaves read
    at3 = t.scan(columns=["id", "m"]).to_arrow().sort_by("id")
    assert at3.column("m").to_pylist() == exp_m


def test_struct_schema_evolution(catalog):
    t = catalog.create_table(
        "sm_ev", Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1)
    t.upsert({"id": np.arange(4, dtype=np.int64), "v": np.zeros(4)})
    t.add_columns([Field("st", ST)])
    t2 = catalog.table("sm_ev")
    t2.upsert({"id": np.array([2], dtype=np.int64), "v": np.array([1.0]),
               "st": [{"a": 5, "b": "new"}]})
    got = t2.scan().to_arrow().sort_by("id").column("st").to_pylist()
    assert got[2] == {"a": 5, "b": "new"}
    assert got[0] is None and got[3] is None


def test_map_string_string(catalog):
    """map<string,string> exercises list<string> leaves on BOTH key and
    value sides."""
    t = catalog.create_table(
        "mss", Schema([Field("id", "int64", False),
                       Field("props", "map<string,string>")]),
        primary_keys=["id"], hash_bucket_num=1)
    rows = [{"x": "1", "y": ""}, None, {}]
    t.upsert({"id": np.arange(3, dtype=np.int64), "props": rows})
    got = t.scan().to_arrow().sort_by("id").column("props").to_pylist()
    assert got == _pairs(rows)


def test_sql_struct_member_access(catalog):
    """SELECT st.a / GROUP BY st.b — Spark-style dotted member access
    through the SQL layer (tensor engine and pandas oracle agree)."""
    import os

    import numpy as np

    from lakesoul_amd.sql import execute_sql

    t = catalog.create_table(
        "sqlst", Schema([Field("id", "int64", False), Field("st", ST)]),
        primary_keys=["id"], hash_bucket_num=1)
    t.upsert({"id": np.arange(6, dtype=np.int64),
              "st": [None if i == 2 else {"a": i * 10, "b": f"g{i % 2}"}
                     for i in range(6)]})
    df = execute_sql(catalog, "SELECT id, st.a FROM sqlst ORDER BY id")
    assert df.columns.tolist() == ["id", "a"]
    assert df["a"].tolist()[0] == 0 and df["a"].tolist()[3] == 30
    assert df["a"].isna()[2]
    df2 = execute_sql(catalog, "SELECT st.b, sum(st.a) s FROM sqlst "
                               "GROUP BY st.b ORDER BY st.b")
    by_b = {r["b"]: r["s"] for _, r in df2.iterrows()}
    assert by_b["g0"] == 40 and by_b["g1"] == 90
    assert any(r is None for r in df2["b"])  # null structs form a group
    # alias + aggregate on member
    df3 = execute_sql(catalog, "SELECT max(st.a) AS m FROM sqlst")
    assert df3["m"].iloc[0] == 50
    # pandas oracle agrees
    os.environ["LAKESOUL_SQL_PANDAS"] = "1"
    try:
        dfp = execute_sql(catalog, "SELECT id, st.a FROM sqlst ORDER BY id")
    finally:
        del os.environ["LAKESOUL_SQL_PANDAS"]
    assert dfp["a"].fillna(-1).tolist() == df["a"].fillna(-1).tolist()
    # unknown member still errors
    import pytest as _pt

    from lakesoul_amd.sql import SqlError

    with _pt.raises(SqlError):
        execute_sql(catalog, "SELECT st.zzz FROM sqlst")
