"""Filter DSL parsing, evaluation, stats/partition pruning tests."""

import numpy as np
import pytest

from lakesoul_amd.io.filters import Cmp, parse_filter_dsl, resolve_filters
from lakesoul_amd.io.schema import Field, Schema


SCHEMA = Schema(
    [Field("id", "int64", False), Field("v", "float64"), Field("s", "string")]
)


def test_parse_dsl_nested():
    e = parse_filter_dsl("and(gt(id, 10), or(eq(s, 'x'), lteq(v, 1.5)))", SCHEMA)
    assert e.prune_stats({"id": (0, 5)}) is False
    assert e.prune_stats({"id": (0, 50)}) is True


def test_parse_dsl_null():
    e = parse_filter_dsl("eq(v, null)", SCHEMA)
    from lakesoul_amd.io.batch import Batch

    b = Batch.from_dict(
        {"id": np.array([1, 2], np.int64), "v": np.array([1.0, 2.0]), "s": ["a", "b"]},
        SCHEMA,
    )
    import torch

    b.columns["v"].validity = torch.tensor([1, 0], dtype=torch.uint8)
    mask = e.evaluate(b)
    assert mask.tolist() == [False, True]


def test_pk_eq_extraction():
    e = parse_filter_dsl("and(eq(id, 42), gt(v, 0))", SCHEMA)
    assert e.pk_eq_values() == {"id": 42}


def test_table_scan_with_dsl_string(catalog):
    t = catalog.create_table("fdsl", SCHEMA, primary_keys=["id"], hash_bucket_num=2)
    n = 1000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.arange(n, dtype=np.float64), "s": ["a"] * n})
    df = t.scan(filters="and(gteq(v, 10), lt(v, 20))").to_arrow().to_pandas()
    assert sorted(df["id"].tolist()) == list(range(10, 20))


def test_stats_pruning_skips_files(catalog):
    t = catalog.create_table("fstat", SCHEMA, primary_keys=["id"], hash_bucket_num=1)
    # two commits with disjoint id ranges
    t.upsert({"id": np.arange(0, 100, dtype=np.int64), "v": np.zeros(100), "s": ["a"] * 100})
    t.upsert({"id": np.arange(1000, 1100, dtype=np.int64), "v": np.ones(100), "s": ["b"] * 100})
    scan = t.scan(filters=[("id", ">=", 1000)])
    units = scan.plan()
    # the first file (ids 0..99) must be pruned by min/max stats
    assert len(units) == 1 and len(units[0].files) == 1
    df = scan.to_arrow().to_pandas()
    assert len(df) == 100 and df["id"].min() == 1000


def test_partition_pruning_with_filter(catalog):
    t = catalog.create_table(
        "fpart",
        Schema([Field("region", "string", False), Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        range_partitions=["region"],
        hash_bucket_num=1,
    )
    t.upsert(
        {
            "region": ["us"] * 5 + ["eu"] * 5,
            "id": np.arange(10, dtype=np.int64),
            "v": np.zeros(10),
        }
    )
    scan = t.scan(filters=[("region", "==", "eu")])
    units = scan.plan()
    assert len(units) == 1 and units[0].partition_desc == "region=eu"
    df = scan.to_arrow().to_pandas()
    assert sorted(df["id"].tolist()) == [5, 6, 7, 8, 9]


def test_dsl_roundtrip_equivalence(catalog):
    """DSL-parsed filters and tuple filters select identical rows."""
    import numpy as np

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "dslrt", Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    n = 2000
    rng = np.random.default_rng(0)
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": rng.uniform(0, 10, n)})
    cases = [
        ("and(gteq(id, 100), lt(id, 200))",
         [("id", ">=", 100), ("id", "<", 200)]),
        ("or(eq(id, 5), eq(id, 7))", None),
        ("not(lt(v, 5.0))", [("v", ">=", 5.0)]),
    ]
    for dsl, tup in cases:
        a = t.to_pandas(filters=dsl).sort_values("id")["id"].tolist()
        if tup is not None:
            b = t.to_pandas(filters=tup).sort_values("id")["id"].tolist()
            assert a == b, dsl
        assert len(a) > 0
