"""Property-based tests (hypothesis) for the engine's semantic contracts:

- Spark murmur3-32 bit-exactness: vectorized numpy oracle == scalar
  pure-python implementation on arbitrary values (the contract the bucket
  layout depends on — reference utils/hash/spark_murmur3.rs).
- MOR merge (merge_cpu.merge_sorted_files) == a pandas groupby-last model
  on arbitrary upsert histories (reference merge_operator.rs semantics).
- Parquet roundtrip: arbitrary fixed-width/string data survives our
  writer -> pyarrow reader bit-exactly.
"""

import numpy as np
import pandas as pd
import pytest
from hypothesis import given, settings, strategies as st

from lakesoul_amd.utils import murmur3 as m3
from lakesoul_amd.utils.murmur3_np import create_hashes_np

# ---------------------------------------------------------------------- #
# murmur3: numpy oracle == scalar implementation


@settings(max_examples=50, deadline=None)
@given(st.lists(st.integers(-(2**63), 2**63 - 1), min_size=1, max_size=50))
def test_hash_int64_vectorized_matches_scalar(vals):
    arr = np.array(vals, dtype=np.int64)
    vec = create_hashes_np([arr])
    for i, v in enumerate(vals):
        assert vec[i] == np.uint32(m3.hash_int64(v) & 0xFFFFFFFF)


@settings(max_examples=50, deadline=None)
@given(st.lists(st.integers(-(2**31), 2**31 - 1), min_size=1, max_size=50))
def test_hash_int32_vectorized_matches_scalar(vals):
    arr = np.array(vals, dtype=np.int32)
    vec = create_hashes_np([arr])
    for i, v in enumerate(vals):
        assert vec[i] == np.uint32(m3.hash_int32(v) & 0xFFFFFFFF)


@settings(max_examples=50, deadline=None)
@given(st.lists(
    st.floats(allow_nan=False, width=64), min_size=1, max_size=50))
def test_hash_float64_vectorized_matches_scalar(vals):
    arr = np.array(vals, dtype=np.float64)
    vec = create_hashes_np([arr])
    for i, v in enumerate(vals):
        assert vec[i] == np.uint32(m3.hash_float64(v) & 0xFFFFFFFF)


@settings(max_examples=50, deadline=None)
@given(st.lists(st.text(max_size=40), min_size=1, max_size=30))
def test_hash_string_cpp_matches_scalar(vals):
    from lakesoul_amd.ops import cpp

    for s in vals:
        got = cpp().spark_hash_bytes(s.encode(), m3.HASH_SEED)
        assert np.uint32(got & 0xFFFFFFFF) == np.uint32(m3.hash_str(s) & 0xFFFFFFFF)


@settings(max_examples=30, deadline=None)
@given(
    st.lists(st.integers(-(2**63), 2**63 - 1), min_size=1, max_size=30),
    st.lists(st.integers(-(2**31), 2**31 - 1), min_size=1, max_size=30),
)
def test_hash_multi_column_seed_chain(a_vals, b_vals):
    n = min(len(a_vals), len(b_vals))
    a = np.array(a_vals[:n], dtype=np.int64)
    b = np.array(b_vals[:n], dtype=np.int32)
    vec = create_hashes_np([a, b])
    for i in range(n):
        h = m3.hash_int64(int(a[i]), m3.HASH_SEED)
        h = m3.hash_int32(int(b[i]), h)
        assert vec[i] == np.uint32(h & 0xFFFFFFFF)


# ---------------------------------------------------------------------- #
# MOR merge == pandas model


def _np_files_from_history(history):
    """history: list of dict {id: value} in commit order -> per-file
    NpColumns sorted by id (as the writer produces them)."""
    from lakesoul_amd.io.merge_cpu import NpColumn

    files = []
    for h in history:
        ids = np.array(sorted(h.keys()), dtype=np.int64)
        vals = np.array([h[i] for i in ids], dtype=np.float64)
        files.append({
            "id": NpColumn("int64", data=ids),
            "v": NpColumn("float64", data=vals),
        })
    return files


@settings(max_examples=40, deadline=None)
@given(st.lists(
    st.dictionaries(st.integers(0, 40), st.floats(allow_nan=False, width=32),
                    min_size=1, max_size=25),
    min_size=1, max_size=6,
))
def test_merge_use_last_matches_pandas(history):
    from lakesoul_amd.io.merge_cpu import merge_sorted_files

    files = _np_files_from_history(history)
    merged = merge_sorted_files(files, ["id"])
    # pandas model: concat in commit order, keep last per id
    frames = [pd.DataFrame({"id": sorted(h.keys()),
                            "v": [h[i] for i in sorted(h.keys())]})
              for h in history]
    ref = (pd.concat(frames, ignore_index=True)
             .groupby("id", as_index=False).last().sort_values("id"))
    np.testing.assert_array_equal(merged["id"].data, ref["id"].to_numpy())
    np.testing.assert_allclose(merged["v"].data, ref["v"].to_numpy())


@settings(max_examples=30, deadline=None)
@given(st.lists(
    st.dictionaries(st.integers(0, 30), st.integers(-1000, 1000),
                    min_size=1, max_size=20),
    min_size=1, max_size=5,
))
def test_merge_sum_all_matches_pandas(history):
    from lakesoul_amd.io.merge_cpu import NpColumn, merge_sorted_files

    files = []
    for h in history:
        ids = np.array(sorted(h.keys()), dtype=np.int64)
        vals = np.array([h[i] for i in ids], dtype=np.int64)
        files.append({"id": NpColumn("int64", data=ids),
                      "v": NpColumn("int64", data=vals)})
    merged = merge_sorted_files(files, ["id"], merge_ops={"v": "SumAll"})
    frames = [pd.DataFrame({"id": sorted(h.keys()),
                            "v": [h[i] for i in sorted(h.keys())]})
              for h in history]
    ref = (pd.concat(frames, ignore_index=True)
             .groupby("id", as_index=False)["v"].sum().sort_values("id"))
    np.testing.assert_array_equal(merged["id"].data, ref["id"].to_numpy())
    np.testing.assert_array_equal(merged["v"].data, ref["v"].to_numpy())


# ---------------------------------------------------------------------- #
# parquet roundtrip under random data


@settings(max_examples=25, deadline=None)
@given(
    st.lists(st.integers(-(2**63), 2**63 - 1), min_size=1, max_size=200),
    st.lists(st.floats(width=64), min_size=1, max_size=200),
    st.integers(1, 4),
)
def test_parquet_roundtrip_random(tmp_path_factory, i64s, f64s, codec_pick):
    import pyarrow.parquet as pq
    import torch

    from lakesoul_amd.ops import cpp

    n = min(len(i64s), len(f64s))
    a = np.array(i64s[:n], dtype=np.int64)
    b = np.array(f64s[:n], dtype=np.float64)
    d = tmp_path_factory.mktemp("pq")
    path = str(d / "r.parquet")
    codec = {1: 0, 2: 0, 3: 6, 4: 6}[codec_pick]  # uncompressed / zstd
    cpp().write_parquet(
        path, ["a", "b"], ["int64", "float64"],
        [torch.from_numpy(a), torch.from_numpy(b)],
        [None, None], [None, None], [False, False], 97, codec, 1,
    )
    t = pq.read_table(path)
    np.testing.assert_array_equal(t.column("a").to_numpy(), a)
    got_b = t.column("b").to_numpy()
    # NaNs compare elementwise
    mask = np.isnan(b)
    np.testing.assert_array_equal(np.isnan(got_b), mask)
    np.testing.assert_array_equal(got_b[~mask], b[~mask])


@settings(max_examples=20, deadline=None)
@given(st.lists(st.binary(max_size=60), min_size=1, max_size=120))
def test_parquet_string_roundtrip_random(tmp_path_factory, items):
    import pyarrow.parquet as pq
    import torch

    from lakesoul_amd.ops import cpp

    offs = np.zeros(len(items) + 1, dtype=np.int32)
    offs[1:] = np.cumsum([len(e) for e in items])
    bys = np.frombuffer(b"".join(items), dtype=np.uint8).copy()
    d = tmp_path_factory.mktemp("pqs")
    path = str(d / "s.parquet")
    cpp().write_parquet(
        path, ["s"], ["binary"],
        [torch.from_numpy(bys)], [torch.from_numpy(offs)],
        [None], [False], 50, 6, 1,
    )
    t = pq.read_table(path)
    assert t.column("s").to_pylist() == items


# ---------------------------------------------------------------------- #
# SQL parser/tokenizer robustness


@settings(max_examples=150, deadline=None)
@given(st.text(max_size=80))
def test_sql_tokenizer_never_hangs(text):
    """Arbitrary input either tokenizes or raises SqlError — never hangs
    or throws anything else."""
    from lakesoul_amd.sql import SqlError, tokenize

    try:
        tokenize(text)
    except SqlError:
        pass


_ident = st.sampled_from(["id", "price", "qty", "region"])
_lit = st.one_of(st.integers(-100, 100),
                 st.sampled_from(["'east'", "'west'", "3.5"]))
_cmp = st.sampled_from(["=", "!=", "<", "<=", ">", ">="])


@settings(max_examples=60, deadline=None)
@given(
    st.lists(_ident, min_size=1, max_size=3, unique=True),
    st.lists(st.tuples(_ident, _cmp, _lit), min_size=0, max_size=3),
    st.sampled_from(["", "ORDER BY id", "ORDER BY id DESC"]),
    st.integers(0, 20),
)
def test_generated_selects_parse(cols, preds, order, limit):
    """Structured random SELECTs always parse into a well-formed AST."""
    from lakesoul_amd.sql import parse_sql

    sql = "SELECT " + ", ".join(cols) + " FROM orders"
    if preds:
        sql += " WHERE " + " AND ".join(f"{c} {o} {v}" for c, o, v in preds)
    if order:
        sql += " " + order
    if limit:
        sql += f" LIMIT {limit}"
    kind, q = parse_sql(sql)
    assert kind == "select"
    assert [i.name for i in q.items] == cols
    assert q.limit == (limit or None)


# ---------------------------------------------------------------------- #
# SQL executor vs pandas oracle on generated queries


@pytest.fixture(scope="module")
def sql_oracle_table(tmp_path_factory):
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.io.schema import Field, Schema

    d = tmp_path_factory.mktemp("sqlprop")
    cat = LakeSoulCatalog(MetaClient(SqliteMetaStore(str(d / "m.db"))),
                          warehouse=str(d / "wh"))
    t = cat.create_table(
        "props",
        Schema([Field("id", "int64", False), Field("price", "float64"),
                Field("qty", "int64"), Field("region", "string")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    rng = np.random.default_rng(0)
    n = 500
    df = pd.DataFrame({
        "id": np.arange(n, dtype=np.int64),
        "price": rng.uniform(0, 100, n).round(2),
        "qty": rng.integers(0, 10, n),
        "region": [["east", "west"][i % 2] for i in range(n)],
    })
    t.upsert({c: df[c].to_numpy() if c != "region" else df[c].tolist()
              for c in df.columns})
    return cat, df


_pcol = st.sampled_from(["id", "price", "qty"])
_pop = st.sampled_from([("=", "=="), ("!=", "!="), ("<", "<"),
                        ("<=", "<="), (">", ">"), (">=", ">=")])
_pval = st.integers(0, 80)


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(_pcol, _pop, _pval), min_size=1, max_size=3))
def test_sql_where_matches_pandas(sql_oracle_table, preds):
    from lakesoul_amd.sql import execute_sql

    cat, df = sql_oracle_table
    where = " AND ".join(f"{c} {sqlop} {v}" for c, (sqlop, _), v in preds)
    got = execute_sql(cat, f"SELECT id FROM props WHERE {where} ORDER BY id")
    mask = pd.Series(True, index=df.index)
    for c, (_, pdop), v in preds:
        mask &= eval(f"df[c] {pdop} v")
    expect = df[mask]["id"].sort_values().tolist()
    assert got["id"].tolist() == expect


@settings(max_examples=25, deadline=None)
@given(
    st.lists(
        st.tuples(
            st.dictionaries(st.integers(0, 25), st.integers(-99, 99),
                            min_size=1, max_size=15),
            st.booleans(),   # include column b in this commit?
        ),
        min_size=1, max_size=5,
    )
)
def test_merge_partial_columns_matches_model(history):
    """Partial-column upserts (schema-evolution writes): UseLast must pick
    the newest file that HAS the column — modeled per key/column."""
    from lakesoul_amd.io.merge_cpu import NpColumn, merge_sorted_files

    files, present = [], []
    model_a, model_b = {}, {}
    for h, with_b in history:
        ids = np.array(sorted(h.keys()), dtype=np.int64)
        a_vals = np.array([h[i] for i in ids], dtype=np.int64)
        cols = {"id": NpColumn("int64", data=ids),
                "a": NpColumn("int64", data=a_vals)}
        pres = {"id", "a"}
        if with_b:
            cols["b"] = NpColumn("int64", data=a_vals * 2)
            pres.add("b")
        else:
            # alignment requires null-filled placeholder
            cols["b"] = NpColumn("int64", data=np.zeros(len(ids), np.int64),
                                 validity=np.zeros(len(ids), np.uint8))
        files.append(cols)
        present.append(pres)
        for i in ids:
            model_a[int(i)] = h[int(i)]
            if with_b:
                model_b[int(i)] = h[int(i)] * 2
    merged = merge_sorted_files(files, ["id"], present=present)
    got_ids = merged["id"].data.tolist()
    assert got_ids == sorted(model_a)
    np.testing.assert_array_equal(
        merged["a"].data, [model_a[i] for i in got_ids])
    bv = merged["b"].validity
    for pos, i in enumerate(got_ids):
        if i in model_b:
            assert bv is None or bv[pos]
            assert merged["b"].data[pos] == model_b[i]
        else:
            assert bv is not None and not bv[pos]


@settings(max_examples=40, deadline=None)
@given(st.lists(st.tuples(
    st.sampled_from(["a", "b", "c", "d_1", "x"]),
    st.sampled_from(["int64", "int32", "float64", "float32", "string",
                     "binary", "date32", "timestamp[us]", "decimal(10,2)",
                     "bool", "int8", "int16"]),
    st.booleans(),
), min_size=1, max_size=6, unique_by=lambda t: t[0]))
def test_schema_json_roundtrip(fields):
    """Spark-style schema JSON serde is a lossless roundtrip for every
    supported dtype (table_schema compatibility contract)."""
    from lakesoul_amd.io.schema import (Field, Schema, schema_from_json,
                                        schema_to_json)

    sch = Schema([Field(n, d, nu) for n, d, nu in fields])
    back = schema_from_json(schema_to_json(sch))
    assert back == sch


def test_murmur3_edge_values():
    """Spark hash contract edge cases: -0.0 folds to 0.0, NaN bit
    patterns hash by their bits, int min/max, empty string."""
    from lakesoul_amd.utils import murmur3 as m3
    from lakesoul_amd.utils.murmur3_np import create_hashes_np

    assert m3.hash_float64(-0.0) == m3.hash_float64(0.0)
    assert m3.hash_float32(-0.0) == m3.hash_float32(0.0)
    a = np.array([0.0, -0.0], dtype=np.float64)
    h = create_hashes_np([a])
    assert h[0] == h[1]
    for v in (-(2**63), 2**63 - 1, 0, -1):
        assert create_hashes_np([np.array([v], dtype=np.int64)])[0] == \
            np.uint32(m3.hash_int64(v) & 0xFFFFFFFF)
    assert m3.hash_str("") == m3.hash_bytes(b"")


# ---------------------------------------------------------------------- #
# nested types: arbitrary list<string>/struct/map data survives
# batch -> parquet -> pyarrow AND batch -> arrow -> batch round trips

_str_elem = st.text(
    alphabet=st.characters(min_codepoint=32, max_codepoint=0x2FF),
    max_size=8)


@settings(max_examples=25, deadline=None)
@given(
    tags=st.lists(st.one_of(st.none(), st.lists(_str_elem, max_size=4)),
                  min_size=1, max_size=20),
    structs=st.lists(
        st.one_of(st.none(),
                  st.fixed_dictionaries({"a": st.integers(-2**40, 2**40),
                                         "b": _str_elem})),
        min_size=1, max_size=20),
    maps=st.lists(
        st.one_of(st.none(),
                  st.dictionaries(_str_elem, st.integers(-2**40, 2**40),
                                  max_size=3)),
        min_size=1, max_size=20),
)
def test_nested_types_roundtrip_random(tmp_path_factory, tags, structs, maps):
    import pyarrow.parquet as pq

    from lakesoul_amd.io.batch import Batch
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.io.writer import _write_batch_to_file_local

    n = min(len(tags), len(structs), len(maps))
    tags, structs, maps = tags[:n], structs[:n], maps[:n]
    s = Schema([Field("id", "int64", False), Field("tags", "list<string>"),
                Field("st", "struct<a:int64,b:string>"),
                Field("mp", "map<string,int64>")])
    b = Batch.from_dict({"id": np.arange(n, dtype=np.int64), "tags": tags,
                         "st": structs, "mp": maps}, s)
    # arrow round trip
    t = b.to_arrow()
    b2 = Batch.from_arrow(t, s)
    assert b2.to_arrow().equals(t)
    # parquet -> pyarrow golden reader
    path = str(tmp_path_factory.mktemp("nested") / "n.parquet")
    _write_batch_to_file_local(path, b, "zstd", 1, max(1, n // 2))
    got = pq.read_table(path)
    assert got.column("tags").to_pylist() == t.column("tags").to_pylist()
    assert got.column("st").to_pylist() == t.column("st").to_pylist()
    assert got.column("mp").to_pylist() == t.column("mp").to_pylist()
    # take-permutation preserves rows
    import torch as _t

    perm = np.random.default_rng(0).permutation(n)
    tk = b.take(_t.from_numpy(perm)).to_arrow()
    for cname in ("tags", "st", "mp"):
        ref = t.column(cname).to_pylist()
        assert tk.column(cname).to_pylist() == [ref[i] for i in perm]
