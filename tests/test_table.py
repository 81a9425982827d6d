"""End-to-end table tests on the CPU path: write / upsert / MOR scan /
merge operators / CDC / compaction / time travel / sharding."""

import os

import numpy as np
import pandas as pd
import pytest

from lakesoul_amd.io.schema import Field, Schema


def _mk_pk_table(catalog, name="t", buckets=4, extra=(), props=None):
    fields = [Field("id", "int64", False), Field("v", "float64"), Field("s", "string")]
    fields += list(extra)
    return catalog.create_table(
        name,
        Schema(fields),
        primary_keys=["id"],
        hash_bucket_num=buckets,
        properties=props,
    )


def _df(tbl, **kwargs):
    df = tbl.to_pandas(**kwargs)
    return df.sort_values("id").reset_index(drop=True)


def test_write_scan_roundtrip_no_pk(catalog):
    t = catalog.create_table(
        "plain", Schema([Field("a", "int64"), Field("b", "float64")])
    )
    data = {"a": np.arange(100, dtype=np.int64), "b": np.ones(100)}
    t.write(data)
    df = t.to_pandas()
    assert len(df) == 100
    np.testing.assert_array_equal(np.sort(df["a"].to_numpy()), np.arange(100))
    assert t.scan().count() == 100


def test_upsert_use_last(catalog):
    t = _mk_pk_table(catalog, "up1")
    n = 10000
    rng = np.random.default_rng(0)
    base = {
        "id": np.arange(n, dtype=np.int64),
        "v": rng.normal(size=n),
        "s": [f"s{i}" for i in range(n)],
    }
    t.upsert(base)
    # upsert 2000 overlapping keys + 500 new
    up_ids = np.concatenate([rng.choice(n, 2000, replace=False), np.arange(n, n + 500)])
    up = {
        "id": up_ids.astype(np.int64),
        "v": np.full(len(up_ids), 99.0),
        "s": [f"u{i}" for i in range(len(up_ids))],
    }
    t.upsert(up)

    df = _df(t)
    assert len(df) == n + 500
    # expected via pandas
    expect = pd.concat(
        [pd.DataFrame(base), pd.DataFrame(up)], ignore_index=True
    ).groupby("id", as_index=False).last().sort_values("id").reset_index(drop=True)
    np.testing.assert_array_equal(df["id"].to_numpy(), expect["id"].to_numpy())
    np.testing.assert_allclose(df["v"].to_numpy(), expect["v"].to_numpy())
    assert df["s"].tolist() == expect["s"].tolist()


def test_multiple_upserts_many_files(catalog):
    t = _mk_pk_table(catalog, "up2", buckets=2)
    n = 2000
    state = {}
    rng = np.random.default_rng(1)
    ids0 = np.arange(n, dtype=np.int64)
    t.upsert({"id": ids0, "v": np.zeros(n), "s": ["a"] * n})
    for i in ids0:
        state[int(i)] = 0.0
    for it in range(5):
        ids = rng.choice(n, 300, replace=False).astype(np.int64)
        vals = np.full(300, float(it + 1))
        t.upsert({"id": ids, "v": vals, "s": [f"it{it}"] * 300})
        for i in ids:
            state[int(i)] = float(it + 1)
    df = _df(t)
    assert len(df) == n
    expect_v = np.array([state[i] for i in df["id"]])
    np.testing.assert_allclose(df["v"].to_numpy(), expect_v)


def test_merge_operator_sum_all(catalog):
    t = catalog.create_table(
        "sums",
        Schema([Field("id", "int64", False), Field("cnt", "int64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
        properties={"merge_op.cnt": "SumAll"},
    )
    t.upsert({"id": np.array([1, 2, 3], dtype=np.int64), "cnt": np.array([10, 20, 30], dtype=np.int64)})
    t.upsert({"id": np.array([2, 3, 4], dtype=np.int64), "cnt": np.array([1, 2, 3], dtype=np.int64)})
    df = _df(t)
    got = dict(zip(df["id"], df["cnt"]))
    assert got == {1: 10, 2: 21, 3: 32, 4: 3}


def test_merge_operator_use_last_not_null(catalog):
    t = catalog.create_table(
        "ulnn",
        Schema([Field("id", "int64", False), Field("x", "float64")]),
        primary_keys=["id"],
        properties={"merge_op.x": "UseLastNotNull"},
    )
    t.upsert({"id": np.array([1, 2], dtype=np.int64), "x": np.array([1.5, 2.5])})
    import pyarrow as pa

    t.upsert(pa.table({"id": pa.array([1, 2], pa.int64()), "x": pa.array([None, 7.5], pa.float64())}))
    df = _df(t)
    assert df["x"].tolist() == [1.5, 7.5]


def test_partial_column_upsert(catalog):
    t = catalog.create_table(
        "partial",
        Schema([Field("id", "int64", False), Field("a", "float64"), Field("b", "float64")]),
        primary_keys=["id"],
    )
    t.upsert({"id": np.array([1, 2], dtype=np.int64), "a": np.array([1.0, 2.0]), "b": np.array([10.0, 20.0])})
    # partial upsert: only column a — b keeps old values
    t.upsert({"id": np.array([1, 2], dtype=np.int64), "a": np.array([5.0, 6.0])})
    df = _df(t)
    assert df["a"].tolist() == [5.0, 6.0]
    assert df["b"].tolist() == [10.0, 20.0]


def test_cdc_delete_rows_dropped(catalog):
    t = catalog.create_table(
        "cdc",
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("rowKinds", "string")]),
        primary_keys=["id"],
        properties={"lakesoul_cdc_change_column": "rowKinds"},
    )
    t.upsert({"id": np.array([1, 2, 3], dtype=np.int64), "v": np.ones(3), "rowKinds": ["insert"] * 3})
    t.upsert({"id": np.array([2], dtype=np.int64), "v": np.array([0.0]), "rowKinds": ["delete"]})
    df = _df(t)
    assert df["id"].tolist() == [1, 3]


def test_compaction_preserves_data(catalog, tmp_path):
    t = _mk_pk_table(catalog, "comp", buckets=2)
    n = 3000
    rng = np.random.default_rng(2)
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n), "s": ["x"] * n})
    for it in range(4):
        ids = rng.choice(n, 200, replace=False).astype(np.int64)
        t.upsert({"id": ids, "v": np.full(200, it + 1.0), "s": ["y"] * 200})
    before = _df(t)
    files_before = t.files()
    t.compaction()
    after = _df(t)
    files_after = t.files()
    pd.testing.assert_frame_equal(before, after)
    assert len(files_after) < len(files_before)
    assert all("compactdir" in f.path for f in files_after)
    # scan after further upsert on top of compacted base
    t.upsert({"id": np.array([0], dtype=np.int64), "v": np.array([123.0]), "s": ["z"]})
    df = _df(t)
    assert df.loc[df["id"] == 0, "v"].iloc[0] == 123.0


def test_time_travel(catalog):
    t = _mk_pk_table(catalog, "tt", buckets=1)
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([1.0]), "s": ["a"]})
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([2.0]), "s": ["b"]})
    assert _df(t)["v"].tolist() == [2.0]
    assert _df(t, version=0)["v"].tolist() == [1.0]


def test_bucket_pruning_point_lookup(catalog):
    t = _mk_pk_table(catalog, "prune", buckets=8)
    n = 5000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.arange(n, dtype=np.float64), "s": ["x"] * n})
    scan = t.scan(filters=[("id", "==", 1234)])
    units = scan.plan()
    assert len(units) == 1  # only the matching bucket
    df = scan.to_arrow().to_pandas()
    assert len(df) == 1 and df["v"].iloc[0] == 1234.0


def test_shard_disjoint_cover(catalog):
    t = _mk_pk_table(catalog, "shard", buckets=8)
    n = 4000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n), "s": ["x"] * n})
    ids = []
    for rank in range(4):
        df = t.scan().shard(rank, 4).to_arrow().to_pandas()
        ids.append(df["id"].to_numpy())
    allids = np.sort(np.concatenate(ids))
    np.testing.assert_array_equal(allids, np.arange(n))


def test_string_pk(catalog):
    t = catalog.create_table(
        "spk",
        Schema([Field("k", "string", False), Field("v", "int64")]),
        primary_keys=["k"],
        hash_bucket_num=4,
    )
    t.upsert({"k": [f"key{i}" for i in range(100)], "v": np.arange(100, dtype=np.int64)})
    t.upsert({"k": ["key5", "key50", "zzz"], "v": np.array([500, 5000, 1], dtype=np.int64)})
    df = t.to_pandas().sort_values("k").reset_index(drop=True)
    assert len(df) == 101
    got = dict(zip(df["k"], df["v"]))
    assert got["key5"] == 500 and got["key50"] == 5000 and got["zzz"] == 1
    assert got["key6"] == 6


def test_multi_pk(catalog):
    t = catalog.create_table(
        "mpk",
        Schema([Field("a", "int64", False), Field("b", "int32", False), Field("v", "float64")]),
        primary_keys=["a", "b"],
        hash_bucket_num=4,
    )
    t.upsert({"a": np.array([1, 1, 2], dtype=np.int64), "b": np.array([1, 2, 1], dtype=np.int32), "v": np.array([1.0, 2.0, 3.0])})
    t.upsert({"a": np.array([1], dtype=np.int64), "b": np.array([2], dtype=np.int32), "v": np.array([9.0])})
    df = t.to_pandas().sort_values(["a", "b"]).reset_index(drop=True)
    assert df["v"].tolist() == [1.0, 9.0, 3.0]


def test_range_partitions(catalog):
    t = catalog.create_table(
        "ranged",
        Schema([Field("dt", "string", False), Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        range_partitions=["dt"],
        hash_bucket_num=2,
    )
    t.upsert(
        {
            "dt": ["2024-01-01"] * 3 + ["2024-01-02"] * 2,
            "id": np.arange(5, dtype=np.int64),
            "v": np.arange(5, dtype=np.float64),
        }
    )
    descs = t.partition_descs()
    assert sorted(descs) == ["dt=2024-01-01", "dt=2024-01-02"]
    df = t.to_pandas().sort_values("id")
    assert len(df) == 5
    assert df["dt"].tolist() == ["2024-01-01"] * 3 + ["2024-01-02"] * 2
    # partition-pruned scan
    df1 = t.scan(partitions=["dt=2024-01-02"]).to_arrow().to_pandas()
    assert sorted(df1["id"].tolist()) == [3, 4]


def test_bucketless_file_not_pruned(catalog):
    """Files without the part-*_NNNN bucket suffix belong to no known
    bucket and must survive PK point-filter bucket pruning."""
    import shutil

    t = _mk_pk_table(catalog, "bless", buckets=4)
    t.upsert({"id": np.arange(50, dtype=np.int64),
              "v": np.arange(50, dtype=np.float64), "s": ["x"] * 50})
    # rename one committed file to drop the bucket suffix, keep metadata in sync
    store = t.client.store
    files = t.client.files_for_partition(t.table_id, "-5")
    src = files[0].path
    dst = os.path.join(os.path.dirname(src), "imported-foreign-file.parquet")
    shutil.move(src, dst)
    with store._conn() as con:
        con.execute(
            "UPDATE data_commit_info SET file_ops = REPLACE(file_ops, ?, ?)",
            (src, dst))
    # a full-PK point filter must still see rows from the renamed file
    for pk in range(50):
        got = t.scan(filters=[("id", "==", pk)]).to_arrow().to_pandas()
        if len(got) == 0:
            raise AssertionError(f"pk {pk} lost after bucket-suffix removal")


def test_range_partition_special_values(catalog):
    """Partition values containing ',', '=', empty string and NULL must
    round-trip through the partition_desc sentinel encoding (reference
    helpers/mod.rs:206-221, constant.rs:18-21)."""
    t = catalog.create_table(
        "ranged_special",
        Schema([Field("city", "string"), Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        range_partitions=["city"],
        hash_bucket_num=2,
    )
    cities = ["a,b", "x=y", "", None, "plain"]
    t.upsert({
        "city": cities,
        "id": np.arange(5, dtype=np.int64),
        "v": np.arange(5, dtype=np.float64),
    })
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == 5
    got = df["city"].tolist()
    assert got[0] == "a,b" and got[1] == "x=y" and got[2] == ""
    assert got[3] is None or (isinstance(got[3], float) and np.isnan(got[3]))
    assert got[4] == "plain"
    # 5 distinct partitions despite the tricky values
    assert len(t.partition_descs()) == 5
    # filter on the decoded value still prunes/matches correctly
    df1 = t.scan(filters="eq(city, 'a,b')").to_arrow().to_pandas()
    assert df1["id"].tolist() == [0]
    # NULL partition: IS NULL finds it, equality prunes it
    df2 = t.scan(filters="eq(city, null)").to_arrow().to_pandas()
    assert df2["id"].tolist() == [3]


def test_range_partition_null_numeric(catalog):
    """NULL in a numeric range-partition column round-trips as null."""
    t = catalog.create_table(
        "ranged_nullnum",
        Schema([Field("bucket_no", "int32"), Field("id", "int64", False)]),
        primary_keys=["id"],
        range_partitions=["bucket_no"],
        hash_bucket_num=1,
    )
    import pandas as _pd

    t.upsert(_pd.DataFrame({
        "bucket_no": _pd.array([1, None], dtype="Int32"),
        "id": np.array([0, 1], dtype=np.int64),
    }))
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert df["bucket_no"].tolist()[0] == 1
    assert _pd.isna(df["bucket_no"].tolist()[1])


def test_incremental_read(catalog):
    t = _mk_pk_table(catalog, "incr", buckets=1)
    t.upsert({"id": np.array([1, 2], dtype=np.int64), "v": np.array([1.0, 2.0]), "s": ["a", "b"]})
    t.upsert({"id": np.array([3], dtype=np.int64), "v": np.array([3.0]), "s": ["c"]})
    t.upsert({"id": np.array([4], dtype=np.int64), "v": np.array([4.0]), "s": ["d"]})
    from lakesoul_amd.io.reader import LakeSoulScan

    scan = LakeSoulScan(t, incremental=(0, 2))
    df = scan.to_arrow().to_pandas().sort_values("id")
    assert df["id"].tolist() == [3, 4]


def test_delete_partition(catalog):
    t = _mk_pk_table(catalog, "delp", buckets=1)
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([1.0]), "s": ["a"]})
    t.delete_partition("-5")
    assert t.scan().count() == 0


def test_invalid_file_tolerance(catalog):
    """Files <8 bytes or missing are skipped with a warning
    (reference session.rs:440-450)."""
    import os
    import warnings

    t = _mk_pk_table(catalog, "tol", buckets=1)
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10), "s": ["a"] * 10})
    # register a bogus tiny file + a missing file into the snapshot
    from lakesoul_amd.meta.entities import CommitOp, DataCommitInfo, DataFileOp, FileOp

    bad = os.path.join(t.table_path, "part-bogus000000000_0000.parquet")
    with open(bad, "wb") as f:
        f.write(b"xx")
    missing = os.path.join(t.table_path, "part-gone0000000000_0000.parquet")
    t.client.commit_data_commit_info(
        DataCommitInfo(
            table_id=t.table_id, partition_desc="-5",
            file_ops=[DataFileOp(bad, FileOp.add, 2), DataFileOp(missing, FileOp.add, 2)],
            commit_op=CommitOp.MergeCommit,
        )
    )
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        df = t.to_pandas()
    assert len(df) == 10
    assert any("invalid" in str(x.message) or "missing" in str(x.message) for x in w)


def test_oversized_unit_chunked_merge(catalog, monkeypatch):
    """Buckets over the memory budget merge in PK ranges (chunked spill
    merge) and return exactly the same result as an unrestricted scan."""
    import warnings

    t = _mk_pk_table(catalog, "huge", buckets=1)
    n = 40000
    rng = np.random.default_rng(3)
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": rng.normal(size=n),
              "s": ["a"] * n})
    for it in range(3):
        ids = rng.choice(n, 8000, replace=False).astype(np.int64)
        t.upsert({"id": ids, "v": rng.normal(size=8000), "s": [f"u{it}"] * 8000})
    full = _df(t)
    monkeypatch.setenv("LAKESOUL_MAX_UNIT_BYTES", "200000")
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        chunked = _df(t)
    assert any("chunked" in str(x.message) for x in w)
    pd.testing.assert_frame_equal(full, chunked)


def test_oversized_unit_chunked_merge_string_pk(catalog, monkeypatch):
    """String-PK oversized buckets also merge in PK ranges (lexicographic
    cut points from byte-array row-group stats) — same result as an
    unrestricted scan."""
    import warnings

    t = catalog.create_table(
        "hugestr",
        Schema([Field("k", "string", False), Field("v", "float64")]),
        primary_keys=["k"], hash_bucket_num=1,
    )
    n = 20000
    rng = np.random.default_rng(5)
    keys = [f"x{i:06d}" for i in range(n)]
    t.upsert({"k": keys, "v": rng.normal(size=n)})
    for it in range(3):
        sel = rng.choice(n, 4000, replace=False)
        t.upsert({"k": [keys[i] for i in sel],
                  "v": np.full(4000, float(it + 10))})
    full = t.to_pandas().sort_values("k").reset_index(drop=True)
    monkeypatch.setenv("LAKESOUL_MAX_UNIT_BYTES", "150000")
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        chunked = t.to_pandas().sort_values("k").reset_index(drop=True)
    assert any("chunked" in str(x.message) for x in w), \
        [str(x.message) for x in w]
    pd.testing.assert_frame_equal(full, chunked)


def test_oversized_unit_guard_composite_pk(catalog, monkeypatch):
    """Non-chunkable (composite PK) oversized buckets still fail with the
    actionable error on CPU scans."""
    t = catalog.create_table(
        "hugecomp",
        Schema([Field("k", "string", False), Field("k2", "int64", False),
                Field("v", "float64")]),
        primary_keys=["k", "k2"], hash_bucket_num=1,
    )
    t.upsert({"k": [f"x{i}" for i in range(1000)],
              "k2": np.arange(1000, dtype=np.int64), "v": np.zeros(1000)})
    monkeypatch.setenv("LAKESOUL_MAX_UNIT_BYTES", "1000")
    with pytest.raises(MemoryError, match="hash\\s*buckets|buckets"):
        t.to_pandas()


def test_decimal_column_table(catalog):
    """decimal(p,s) end-to-end: upsert, MOR merge, filters, SQL, SumAll."""
    import decimal

    import pyarrow as pa

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "dec",
        Schema([Field("id", "int64", False), Field("amt", "decimal(12,2)")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    # write via arrow decimals
    t.upsert(pa.table({
        "id": pa.array([1, 2, 3], pa.int64()),
        "amt": pa.array([decimal.Decimal("10.50"), decimal.Decimal("-1.25"),
                         decimal.Decimal("3.00")], pa.decimal128(12, 2)),
    }))
    # upsert overwrite via raw unscaled ints
    t.upsert({"id": np.array([2], dtype=np.int64),
              "amt": np.array([9900], dtype=np.int64)})
    df = _df(t)
    assert df["amt"].tolist() == [decimal.Decimal("10.50"),
                                  decimal.Decimal("99.00"),
                                  decimal.Decimal("3.00")]
    # filter with a logical literal
    got = t.to_pandas(filters=[("amt", ">", 5.0)])
    assert sorted(got["id"]) == [1, 2]
    # SQL over decimal
    from lakesoul_amd.sql import execute_sql

    q = execute_sql(catalog, "SELECT id FROM dec WHERE amt >= 10.5 ORDER BY id")
    assert q["id"].tolist() == [1, 2]


def test_decimal_sum_all_merge(catalog):
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "decsum",
        Schema([Field("id", "int64", False), Field("amt", "decimal(10,2)")]),
        primary_keys=["id"],
        properties={"merge_op.amt": "SumAll"},
    )
    t.upsert({"id": np.array([1, 2], dtype=np.int64),
              "amt": np.array([100, 250], dtype=np.int64)})
    t.upsert({"id": np.array([2, 3], dtype=np.int64),
              "amt": np.array([50, 75], dtype=np.int64)})
    import decimal

    df = _df(t)
    assert df["amt"].tolist() == [decimal.Decimal("1.00"), decimal.Decimal("3.00"),
                                  decimal.Decimal("0.75")]


def test_string_pk_gpu_merge_logic_cpu(catalog):
    """merge_key_order's string LSD path (CPU reference of the chunk-key
    kernel) must equal python sorted() byte order, including tricky
    shared prefixes, embedded NULs and length ties."""
    import torch

    from lakesoul_amd.io.batch import Batch
    from lakesoul_amd.io.merge_gpu import merge_key_order
    from lakesoul_amd.io.schema import Field, Schema

    sch = Schema([Field("k", "string", False)])
    f1 = ["apple", "apple\x00", "commonprefix_aaaa", "commonprefix_aaab", "z"]
    f2 = ["app", "apple", "commonprefix_aaaa0", "commonprefix_aa", "za"]
    b1 = Batch.from_dict({"k": sorted(f1)}, sch)
    b2 = Batch.from_dict({"k": sorted(f2)}, sch)
    cols = [[b1.columns["k"]], [b2.columns["k"]]]
    order, keys, eq = merge_key_order(cols, [5, 5], torch.device("cpu"))
    assert keys is None and eq is not None
    allk = sorted(f1) + sorted(f2)
    got = [allk[i] for i in order.tolist()]
    # expected: stable sort by byte order, ties keep concat (file) order
    expect = [s for s in sorted(got, key=lambda x: x.encode())]
    assert [g.encode() for g in got] == [e.encode() for e in expect]
    # equal adjacent strings must show equal across ALL eq tensors
    for i in range(len(got) - 1):
        same_eq = all(bool(t[order[i]] == t[order[i + 1]]) for t in eq)
        assert same_eq == (got[i] == got[i + 1]), (got[i], got[i + 1])


def test_chunked_merge_with_merge_ops_and_cdc(catalog, monkeypatch):
    """Chunked spill merge composes with merge operators and CDC."""
    import warnings

    t = catalog.create_table(
        "hugeops",
        Schema([Field("id", "int64", False), Field("cnt", "int64"),
                Field("x", "float64"), Field("rowKinds", "string")]),
        primary_keys=["id"], hash_bucket_num=1,
        properties={"merge_op.cnt": "SumAll",
                    "lakesoul_cdc_change_column": "rowKinds"},
    )
    n = 20000
    rng = np.random.default_rng(5)
    t.upsert({"id": np.arange(n, dtype=np.int64),
              "cnt": np.ones(n, dtype=np.int64),
              "x": rng.normal(size=n), "rowKinds": ["insert"] * n})
    t.upsert({"id": np.arange(0, n, 2, dtype=np.int64),
              "cnt": np.full(n // 2, 10, dtype=np.int64),
              "x": rng.normal(size=n // 2), "rowKinds": ["update"] * (n // 2)})
    t.upsert({"id": np.arange(0, n, 1000, dtype=np.int64),
              "cnt": np.zeros(n // 1000, dtype=np.int64),
              "x": np.zeros(n // 1000), "rowKinds": ["delete"] * (n // 1000)})
    full = _df(t)
    monkeypatch.setenv("LAKESOUL_MAX_UNIT_BYTES", "150000")
    with warnings.catch_warnings(record=True) as w:
        warnings.simplefilter("always")
        chunked = _df(t)
    assert any("chunked" in str(x.message) for x in w), [str(x.message) for x in w]
    pd.testing.assert_frame_equal(full, chunked)
    assert len(full) == n - n // 1000


def test_composite_mixed_pk_merge_order_cpu():
    """merge_key_order generic path: composite (int32, string, int64) PK
    sorts lexicographically per column with stable snapshot ties."""
    import torch

    from lakesoul_amd.io.batch import Batch
    from lakesoul_amd.io.merge_gpu import merge_key_order
    from lakesoul_amd.io.schema import Field, Schema

    sch = Schema([Field("a", "int32", False), Field("b", "string", False),
                  Field("c", "int64", False)])
    rows1 = [(1, "x", 5), (1, "y", 1), (2, "a", 9)]
    rows2 = [(1, "x", 5), (1, "x", 7), (2, "a", 2)]

    def mk(rows):
        return Batch.from_dict({
            "a": np.array([r[0] for r in rows], dtype=np.int32),
            "b": [r[1] for r in rows],
            "c": np.array([r[2] for r in rows], dtype=np.int64),
        }, sch)

    b1, b2 = mk(rows1), mk(rows2)
    cols = [[b1.columns[k] for k in ("a", "b", "c")],
            [b2.columns[k] for k in ("a", "b", "c")]]
    order, keys, eq = merge_key_order(cols, [3, 3], torch.device("cpu"))
    allrows = rows1 + rows2
    got = [allrows[i] for i in order.tolist()]
    assert got == sorted(got)  # python tuple order == our LSD order
    # stability: equal (1,'x',5) rows keep file order (row 0 before row 3)
    i0, i3 = order.tolist().index(0), order.tolist().index(3)
    assert i0 < i3


def test_mor_with_foreign_written_delta(catalog, tmp_path):
    """A delta file written by pyarrow (dict-encoded, snappy) merges
    correctly with our zstd PLAIN files — mixed-codec MOR units."""
    import pyarrow as pa
    import pyarrow.parquet as pq

    from lakesoul_amd.meta.entities import (CommitOp, DataCommitInfo,
                                            DataFileOp, FileOp)

    t = _mk_pk_table(catalog, "foreign", buckets=1)
    n = 5000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n),
              "s": ["ours"] * n})
    # foreign delta: ids 0..999 overwritten, written by pyarrow with
    # dictionary + snappy (sorted by pk as the contract requires)
    foreign = str(tmp_path / "part-foreign0000000_0000.parquet")
    pq.write_table(pa.table({
        "id": pa.array(np.arange(1000, dtype=np.int64)),
        "v": pa.array(np.full(1000, 7.5)),
        "s": pa.array(["theirs"] * 1000),
    }), foreign, use_dictionary=True, compression="snappy")
    t.client.commit_data_commit_info(DataCommitInfo(
        table_id=t.table_id, partition_desc="-5",
        file_ops=[DataFileOp(foreign, FileOp.add, os.path.getsize(foreign))],
        commit_op=CommitOp.MergeCommit,
    ))
    df = _df(t)
    assert len(df) == n
    assert (df[df.id < 1000]["s"] == "theirs").all()
    assert (df[df.id >= 1000]["s"] == "ours").all()
    assert (df[df.id < 1000]["v"] == 7.5).all()


def test_merge_operator_sum_last(catalog):
    """SumLast: last value per FILE per key, summed across files —
    differs from SumAll exactly when one file holds duplicate PKs."""
    t = catalog.create_table(
        "sumlast",
        Schema([Field("id", "int64", False), Field("cnt", "int64")]),
        primary_keys=["id"], hash_bucket_num=1,
        properties={"merge_op.cnt": "SumLast"},
    )
    # one batch with a duplicated key: file keeps both rows (stable sort)
    t.upsert({"id": np.array([1, 1, 2], dtype=np.int64),
              "cnt": np.array([10, 20, 5], dtype=np.int64)})
    t.upsert({"id": np.array([1, 2], dtype=np.int64),
              "cnt": np.array([100, 50], dtype=np.int64)})
    df = _df(t)
    got = dict(zip(df["id"], df["cnt"]))
    # key 1: file1 last=20, file2 last=100 -> 120 (SumAll would give 130)
    assert got == {1: 120, 2: 55}


def test_scan_batch_size_slicing(catalog):
    """scan(batch_size=n) yields bounded batches covering all rows once
    (reference: LakeSoulIOConfig batch_size, config/mod.rs)."""
    t = _mk_pk_table(catalog, "bsz", buckets=2)
    n = 5000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n),
              "s": [f"x{i}" for i in range(n)]})
    got = []
    sizes = []
    for b in t.scan(batch_size=700).iter_batches():
        sizes.append(b.num_rows)
        got.extend(b.columns["id"].data.tolist())
    assert max(sizes) <= 700
    assert sorted(got) == list(range(n))
