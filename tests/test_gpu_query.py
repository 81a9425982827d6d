"""GPU tests for the tensor query engine + SQL on cuda."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def test_engine_groupby_join_sort_on_gpu(dev):
    from lakesoul_amd.io.batch import Batch
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.query.engine import (
        groupby_agg, join_batches, sort_indices)

    rng = np.random.default_rng(0)
    n = 200_000
    schema = Schema([Field("g", "string"), Field("k", "int64", False),
                     Field("v", "float64")])
    b = Batch.from_dict({
        "g": [f"grp{i % 37}" for i in range(n)],
        "k": rng.integers(0, 1000, n),
        "v": rng.normal(size=n),
    }, schema).to_device("cuda")
    out = groupby_agg(b, ["g"], [("count", None, "n", False),
                                 ("sum", "v", "sv", False),
                                 ("max", "k", "mk", False)])
    df = out.to_arrow().to_pandas().sort_values("g").reset_index(drop=True)
    ref = (b.to_device("cpu").to_arrow().to_pandas()
           .groupby("g", as_index=False)
           .agg(n=("g", "size"), sv=("v", "sum"), mk=("k", "max"))
           .sort_values("g").reset_index(drop=True))
    assert df["g"].tolist() == ref["g"].tolist()
    np.testing.assert_allclose(df["sv"].to_numpy(), ref["sv"].to_numpy(), rtol=1e-9)
    np.testing.assert_array_equal(df["n"].to_numpy(), ref["n"].to_numpy())
    # join on gpu
    rschema = Schema([Field("k2", "int64", False), Field("w", "float64")])
    r = Batch.from_dict({"k2": np.arange(1000, dtype=np.int64),
                         "w": np.ones(1000)}, rschema).to_device("cuda")
    joined = join_batches(b, r, ["k"], ["k2"], "inner")
    assert joined.num_rows == n
    # sort on gpu incl. strings
    idx = sort_indices(b, [("g", True), ("v", False)])
    assert idx.device.type == "cuda"
    got = b.take(idx).to_device("cpu").to_arrow().to_pandas()
    assert got["g"].is_monotonic_increasing is True or (
        got["g"].tolist() == sorted(got["g"].tolist()))


def test_sql_on_gpu_device(dev, tmp_path):
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.sql import execute_sql

    catalog = LakeSoulCatalog(
        MetaClient(SqliteMetaStore(str(tmp_path / "meta.db"))),
        warehouse=str(tmp_path / "wh"))
    t = catalog.create_table(
        "g", Schema([Field("id", "int64", False), Field("v", "float64"),
                     Field("tag", "string")]),
        primary_keys=["id"], hash_bucket_num=4)
    n = 100_000
    rng = np.random.default_rng(1)
    t.upsert({"id": np.arange(n, dtype=np.int64),
              "v": rng.normal(size=n),
              "tag": [f"t{i % 5}" for i in range(n)]})
    df = execute_sql(
        catalog,
        "SELECT tag, count(*) n, sum(v * 2) s FROM g GROUP BY tag ORDER BY tag",
        device="cuda")
    assert df["tag"].tolist() == [f"t{i}" for i in range(5)]
    assert df["n"].sum() == n
    ref = execute_sql(
        catalog,
        "SELECT tag, count(*) n, sum(v * 2) s FROM g GROUP BY tag ORDER BY tag",
        device="cpu")
    np.testing.assert_allclose(df["s"].to_numpy(), ref["s"].to_numpy(), rtol=1e-9)
