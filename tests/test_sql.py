"""SQL console layer (reference: rust/lakesoul-console, datafusion cli)."""

import numpy as np
import pandas as pd
import pytest

from lakesoul_amd.io.schema import Field, Schema
from lakesoul_amd.sql import SqlError, execute_sql, parse_sql, repl


@pytest.fixture
def sql_table(catalog):
    t = catalog.create_table(
        "orders",
        Schema([
            Field("id", "int64", False),
            Field("price", "float64"),
            Field("qty", "int64"),
            Field("region", "string"),
        ]),
        primary_keys=["id"],
        hash_bucket_num=4,
    )
    n = 1000
    rng = np.random.default_rng(0)
    t.upsert({
        "id": np.arange(n, dtype=np.int64),
        "price": rng.uniform(1, 100, n).round(2),
        "qty": rng.integers(1, 10, n),
        "region": [["east", "west", "north"][i % 3] for i in range(n)],
    })
    return catalog, t


def _ref_df(t):
    return t.to_pandas().sort_values("id").reset_index(drop=True)


def test_select_star_limit(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT * FROM orders ORDER BY id LIMIT 5")
    assert list(df.columns) == ["id", "price", "qty", "region"]
    assert df["id"].tolist() == [0, 1, 2, 3, 4]


def test_where_pushdown_and_residual(sql_table):
    cat, t = sql_table
    df = execute_sql(
        cat, "SELECT id, price FROM orders WHERE id >= 10 AND id < 20 AND region = 'east'"
    )
    ref = _ref_df(t)
    ref = ref[(ref.id >= 10) & (ref.id < 20) & (ref.region == "east")]
    assert sorted(df["id"]) == sorted(ref["id"])


def test_in_between_not(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT id FROM orders WHERE id IN (3, 5, 7) ORDER BY id")
    assert df["id"].tolist() == [3, 5, 7]
    df = execute_sql(cat, "SELECT id FROM orders WHERE id BETWEEN 4 AND 6 ORDER BY id")
    assert df["id"].tolist() == [4, 5, 6]
    df = execute_sql(
        cat, "SELECT id FROM orders WHERE id < 5 AND NOT (id = 2 OR id = 3) ORDER BY id"
    )
    assert df["id"].tolist() == [0, 1, 4]


def test_count_star_fast_path(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT count(*) FROM orders")
    assert df.iloc[0, 0] == 1000
    df = execute_sql(cat, "SELECT count(*) AS n FROM orders WHERE id < 100")
    assert df["n"].iloc[0] == 100


def test_aggregates_global(sql_table):
    cat, t = sql_table
    df = execute_sql(
        cat, "SELECT sum(qty) total, min(price) lo, max(price) hi, avg(qty) m FROM orders"
    )
    ref = _ref_df(t)
    assert df["total"].iloc[0] == ref["qty"].sum()
    assert df["lo"].iloc[0] == ref["price"].min()
    assert df["hi"].iloc[0] == ref["price"].max()
    assert abs(df["m"].iloc[0] - ref["qty"].mean()) < 1e-9


def test_group_by(sql_table):
    cat, t = sql_table
    df = execute_sql(
        cat,
        "SELECT region, count(*) n, sum(qty) q FROM orders GROUP BY region ORDER BY region",
    )
    ref = (
        _ref_df(t).groupby("region", as_index=False)
        .agg(n=("qty", "size"), q=("qty", "sum"))
        .sort_values("region").reset_index(drop=True)
    )
    assert df["region"].tolist() == ref["region"].tolist()
    assert df["n"].tolist() == ref["n"].tolist()
    assert df["q"].tolist() == ref["q"].tolist()


def test_distinct_and_alias(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT DISTINCT region AS r FROM orders ORDER BY r")
    assert df["r"].tolist() == ["east", "north", "west"]


def test_order_desc(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT id FROM orders ORDER BY id DESC LIMIT 3")
    assert df["id"].tolist() == [999, 998, 997]


def test_show_and_describe(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SHOW TABLES")
    assert "orders" in df["table"].tolist()
    df = execute_sql(cat, "SHOW NAMESPACES")
    assert "default" in df["namespace"].tolist()
    df = execute_sql(cat, "DESCRIBE orders")
    assert df[df.column == "id"]["primary_key"].iloc[0]
    assert df[df.column == "price"]["type"].iloc[0] == "float64"


def test_time_travel_version(sql_table):
    cat, t = sql_table
    t.upsert({"id": np.array([0], dtype=np.int64), "price": np.array([-1.0]),
              "qty": np.array([0], dtype=np.int64), "region": ["x"]})
    now = execute_sql(cat, "SELECT price FROM orders WHERE id = 0")
    old = execute_sql(cat, "SELECT price FROM orders VERSION 0 WHERE id = 0")
    assert now["price"].iloc[0] == -1.0
    assert old["price"].iloc[0] != -1.0


def test_is_null(catalog):
    t = catalog.create_table(
        "nulls", Schema([Field("id", "int64", False), Field("x", "float64")]),
        primary_keys=["id"],
    )
    import pyarrow as pa

    t.upsert(pa.table({"id": pa.array([1, 2, 3], pa.int64()),
                       "x": pa.array([1.0, None, 3.0], pa.float64())}))
    df = execute_sql(catalog, "SELECT id FROM nulls WHERE x IS NULL")
    assert df["id"].tolist() == [2]
    df = execute_sql(catalog, "SELECT id FROM nulls WHERE x IS NOT NULL ORDER BY id")
    assert df["id"].tolist() == [1, 3]


def test_errors(sql_table):
    cat, t = sql_table
    with pytest.raises(SqlError):
        execute_sql(cat, "SELECT nosuch FROM orders")
    with pytest.raises(SqlError):
        execute_sql(cat, "DELETE FROM orders")
    with pytest.raises(SqlError):
        execute_sql(cat, "SELECT id, sum(qty) FROM orders")  # id not grouped
    with pytest.raises(SqlError):
        execute_sql(cat, "SELECT id FROM orders WHERE id ~ 3")


def test_parse_shapes():
    kind, q = parse_sql("select a, b from ns1.t where a = 1 or b = 'x' limit 3;")
    assert kind == "select" and q.namespace == "ns1" and q.table == "t"
    assert q.limit == 3
    assert [i.name for i in q.items] == ["a", "b"]


def test_repl_loop(sql_table, capsys):
    cat, t = sql_table
    lines = iter(["SELECT count(*) FROM orders", "bogus(", "\\q"])
    outs = []
    repl(cat, input_fn=lambda _: next(lines), print_fn=outs.append)
    joined = "\n".join(str(o) for o in outs)
    assert "1000" in joined
    assert "error:" in joined
