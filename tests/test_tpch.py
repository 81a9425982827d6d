"""TPC-H q1/q3/q6 through execute_sql cross-checked against the bespoke
tensor paths (benchmarks/tpch.py) — the SQL layer must produce the same
numbers as the hand-built pipelines it benchmarks."""

import os
import sys

import numpy as np
import pytest
import torch

sys.path.insert(0, os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "benchmarks"))


@pytest.fixture(scope="module")
def tpch_tables(tmp_path_factory):
    td = tmp_path_factory.mktemp("tpch")
    os.environ["LAKESOUL_META_DB"] = str(td / "meta.db")
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog

    import tpch as T

    catalog = LakeSoulCatalog(
        MetaClient(SqliteMetaStore(os.environ["LAKESOUL_META_DB"])),
        warehouse=str(td / "wh"))
    t = T.make_lineitem(catalog, 0.01, "cpu")
    t_orders = T.make_orders(catalog, 0.01, "cpu")
    return catalog, t, t_orders, T


def test_q6_sql_matches_bespoke(tpch_tables):
    catalog, t, _, T = tpch_tables
    bespoke = T.q6(t, "cpu")
    df = T.q6_sql(catalog, "cpu")
    assert abs(df["revenue"].iloc[0] - bespoke) < 1e-6 * max(1.0, abs(bespoke))


def test_q1_sql_matches_bespoke(tpch_tables):
    catalog, t, _, T = tpch_tables
    sums, counts = T.q1lite(t, "cpu")  # (6, 4) sums, (6,) counts
    df = T.q1_sql(catalog, "cpu")
    # map rows of df (returnflag, linestatus) to group id rf*2+ls
    for _, row in df.iterrows():
        g = int(row["l_returnflag"]) * 2 + int(row["l_linestatus"])
        np.testing.assert_allclose(row["sum_qty"], float(sums[g, 0]), rtol=1e-9)
        np.testing.assert_allclose(row["sum_base_price"], float(sums[g, 1]), rtol=1e-9)
        np.testing.assert_allclose(row["sum_disc_price"], float(sums[g, 2]), rtol=1e-9)
        np.testing.assert_allclose(row["sum_charge"], float(sums[g, 3]), rtol=1e-9)
        assert row["count_order"] == int(counts[g])
        np.testing.assert_allclose(row["avg_qty"],
                                   float(sums[g, 0]) / float(counts[g]), rtol=1e-9)
    assert len(df) == 6


def test_q3_sql_matches_bespoke(tpch_tables):
    catalog, t, t_orders, T = tpch_tables
    bespoke_top10_sum = T.q3lite(t, t_orders, "cpu")
    df = T.q3_sql(catalog, "cpu")
    assert len(df) <= 10
    np.testing.assert_allclose(df["revenue"].sum(), bespoke_top10_sum,
                               rtol=1e-9)
    # descending revenue order
    rv = df["revenue"].to_numpy()
    assert (rv[:-1] >= rv[1:]).all()
