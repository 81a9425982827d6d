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
        execute_sql(cat, "GRANT SELECT ON orders TO bob")
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


@pytest.fixture
def join_tables(catalog):
    from lakesoul_amd.io.schema import Field, Schema

    o = catalog.create_table(
        "jorders",
        Schema([Field("oid", "int64", False), Field("cust", "int64"),
                Field("amount", "float64")]),
        primary_keys=["oid"], hash_bucket_num=2,
    )
    o.upsert({"oid": np.arange(20, dtype=np.int64),
              "cust": np.arange(20, dtype=np.int64) % 5,
              "amount": np.arange(20, dtype=np.float64) * 10})
    c = catalog.create_table(
        "jcust",
        Schema([Field("cid", "int64", False), Field("name", "string")]),
        primary_keys=["cid"], hash_bucket_num=1,
    )
    c.upsert({"cid": np.arange(4, dtype=np.int64),
              "name": [f"cust{i}" for i in range(4)]})  # cust 4 has no row
    return catalog


def test_inner_join(join_tables):
    df = execute_sql(join_tables,
        "SELECT o.oid, c.name, o.amount FROM jorders o "
        "JOIN jcust c ON o.cust = c.cid ORDER BY o.oid")
    assert len(df) == 16  # cust 4 rows dropped
    assert list(df.columns) == ["oid", "name", "amount"]
    assert df[df.oid == 0]["name"].iloc[0] == "cust0"


def test_left_join_and_where(join_tables):
    df = execute_sql(join_tables,
        "SELECT o.oid, c.name FROM jorders o LEFT JOIN jcust c "
        "ON o.cust = c.cid WHERE o.amount > 50 ORDER BY o.oid")
    assert len(df) == 14  # oids 6..19
    assert df[df.oid == 9]["name"].isna().iloc[0]  # cust 4 unmatched


def test_join_group_by(join_tables):
    df = execute_sql(join_tables,
        "SELECT c.name, sum(o.amount) total FROM jorders o "
        "JOIN jcust c ON o.cust = c.cid GROUP BY c.name ORDER BY c.name")
    # cust0: oids 0,5,10,15 -> 0+50+100+150 = 300
    assert df[df.name == "cust0"]["total"].iloc[0] == 300.0


def test_insert_values_and_select(catalog):
    from lakesoul_amd.io.schema import Field, Schema

    catalog.create_table(
        "ins", Schema([Field("id", "int64", False), Field("v", "float64"),
                       Field("s", "string")]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    r = execute_sql(catalog, "INSERT INTO ins VALUES (1, 1.5, 'a'), (2, 2.5, 'b')")
    assert r["rows_inserted"].iloc[0] == 2
    df = execute_sql(catalog, "SELECT * FROM ins ORDER BY id")
    assert df["s"].tolist() == ["a", "b"]
    # upsert semantics via INSERT on PK table
    execute_sql(catalog, "INSERT INTO ins VALUES (2, 9.0, 'z')")
    df = execute_sql(catalog, "SELECT v FROM ins WHERE id = 2")
    assert df["v"].iloc[0] == 9.0
    # INSERT ... SELECT into a second table
    catalog.create_table(
        "ins2", Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    r = execute_sql(catalog, "INSERT INTO ins2 (id, v) SELECT id, v FROM ins")
    assert r["rows_inserted"].iloc[0] == 2
    assert execute_sql(catalog, "SELECT count(*) n FROM ins2")["n"].iloc[0] == 2


def test_single_table_qualifiers(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT orders.id FROM orders WHERE orders.id < 3 ORDER BY orders.id")
    assert df["id"].tolist() == [0, 1, 2]
    df = execute_sql(cat, "SELECT o.id FROM orders o WHERE o.id = 5")
    assert df["id"].tolist() == [5]
    with pytest.raises(SqlError):
        execute_sql(cat, "SELECT x.id FROM orders o")


def test_update_and_delete_statements(catalog):
    catalog_tables = catalog
    from lakesoul_amd.io.schema import Field, Schema

    catalog.create_table(
        "dml", Schema([Field("id", "int64", False), Field("v", "float64"),
                       Field("tag", "string")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    execute_sql(catalog, "INSERT INTO dml VALUES (1, 1.0, 'a'), (2, 2.0, 'b'), (3, 3.0, 'a')")
    r = execute_sql(catalog, "UPDATE dml SET v = 99.0, tag = 'z' WHERE id >= 2")
    assert r["rows_updated"].iloc[0] == 2
    df = execute_sql(catalog, "SELECT id, v, tag FROM dml ORDER BY id")
    assert df["v"].tolist() == [1.0, 99.0, 99.0]
    assert df["tag"].tolist() == ["a", "z", "z"]
    r = execute_sql(catalog, "DELETE FROM dml WHERE tag = 'z'")
    assert r["rows_deleted"].iloc[0] == 2
    df = execute_sql(catalog, "SELECT count(*) n FROM dml")
    assert df["n"].iloc[0] == 1


def test_having_offset(sql_table):
    cat, t = sql_table
    df = execute_sql(cat,
        "SELECT region, count(*) n FROM orders GROUP BY region "
        "HAVING n > 333 ORDER BY region")
    assert (df["n"] > 333).all()
    df = execute_sql(cat, "SELECT id FROM orders ORDER BY id LIMIT 3 OFFSET 5")
    assert df["id"].tolist() == [5, 6, 7]


def test_count_distinct(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SELECT count(DISTINCT region) r FROM orders")
    assert df["r"].iloc[0] == 3
    df = execute_sql(cat,
        "SELECT region, count(DISTINCT qty) q FROM orders GROUP BY region ORDER BY region")
    assert (df["q"] <= 9).all() and (df["q"] >= 1).all()


def test_timestamp_time_travel(sql_table):
    import time as _time

    cat, t = sql_table
    _time.sleep(0.01)
    ts_between = int(_time.time() * 1000)
    _time.sleep(0.01)
    t.upsert({"id": np.array([0], dtype=np.int64), "price": np.array([-9.0]),
              "qty": np.array([0], dtype=np.int64), "region": ["x"]})
    old = execute_sql(cat, f"SELECT price FROM orders TIMESTAMP {ts_between} WHERE id = 0")
    now = execute_sql(cat, "SELECT price FROM orders WHERE id = 0")
    assert now["price"].iloc[0] == -9.0
    assert old["price"].iloc[0] != -9.0


def test_in_subquery_and_scalar_subquery(join_tables):
    df = execute_sql(join_tables,
        "SELECT oid FROM jorders WHERE cust IN "
        "(SELECT cid FROM jcust WHERE name = 'cust1') ORDER BY oid")
    assert df["oid"].tolist() == [1, 6, 11, 16]
    df = execute_sql(join_tables,
        "SELECT count(*) n FROM jorders WHERE amount > "
        "(SELECT avg(amount) m FROM jorders)")
    assert df["n"].iloc[0] == 10  # amounts 0..190, avg 95 -> 10 above
    df = execute_sql(join_tables,
        "SELECT oid FROM jorders WHERE cust NOT IN (SELECT cid FROM jcust)"
        " ORDER BY oid")
    assert df["oid"].tolist() == [4, 9, 14, 19]  # cust 4 has no row


def test_explain(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "EXPLAIN SELECT id FROM orders WHERE id = 7")
    assert df["scan_units"].iloc[0] == 1  # bucket pruning visible
    assert "id" in df["pushdown"].iloc[0]
    df2 = execute_sql(cat, "EXPLAIN SELECT * FROM orders")
    assert df2["scan_units"].iloc[0] == 4


def test_create_and_drop_table_sql(catalog):
    execute_sql(catalog,
        "CREATE TABLE sales (id BIGINT NOT NULL, amt DECIMAL(10,2), "
        "region VARCHAR(16), d DATE) PRIMARY KEY (id) HASH BUCKETS 2 "
        "PARTITION BY (region)")
    t = catalog.table("sales")
    assert t.primary_keys == ["id"]
    assert t.hash_bucket_num == 2
    assert t.range_keys == ["region"]
    assert t.schema.field("amt").dtype == "decimal(10,2)"
    execute_sql(catalog,
        "INSERT INTO sales VALUES (1, 9.5, 'east', 19000), (2, 1.25, 'west', 19001)")
    df = execute_sql(catalog, "SELECT count(*) n FROM sales")
    assert df["n"].iloc[0] == 2
    import decimal

    amts = execute_sql(catalog, "SELECT id, amt FROM sales ORDER BY id")
    assert amts["amt"].tolist() == [decimal.Decimal("9.50"),
                                    decimal.Decimal("1.25")]
    execute_sql(catalog, "DROP TABLE sales")
    assert not catalog.table_exists("sales")
    execute_sql(catalog, "DROP TABLE IF EXISTS sales")  # no error


def test_arithmetic_expressions(sql_table):
    """Scalar arithmetic in SELECT / aggregates (TPC-H q1/q6 shapes)."""
    cat, t = sql_table
    ref = _ref_df(t)
    # q6 shape: sum of a product under filters
    df = execute_sql(cat, "SELECT sum(price * qty) rev FROM orders "
                          "WHERE qty >= 3 AND price < 50")
    m = (ref["qty"] >= 3) & (ref["price"] < 50)
    assert abs(df["rev"].iloc[0] - (ref[m]["price"] * ref[m]["qty"]).sum()) < 1e-6
    # q1 shape: grouped sums of compound expressions
    df2 = execute_sql(cat, "SELECT region, sum(price * (1 - qty / 100.0)) s1, "
                           "avg(price * 2) a2, count(*) n FROM orders "
                           "GROUP BY region ORDER BY region")
    g = ref.assign(s1=ref["price"] * (1 - ref["qty"] / 100.0),
                   a2=ref["price"] * 2).groupby("region")
    exp = g.agg(s1=("s1", "sum"), a2=("a2", "mean"), n=("region", "size")).reset_index()
    assert df2["region"].tolist() == sorted(exp["region"].tolist())
    np.testing.assert_allclose(df2["s1"].to_numpy(),
                               exp.sort_values("region")["s1"].to_numpy())
    np.testing.assert_allclose(df2["a2"].to_numpy(),
                               exp.sort_values("region")["a2"].to_numpy())
    # computed projection (no aggregates)
    df3 = execute_sql(cat, "SELECT id, price * qty AS total FROM orders "
                           "WHERE id < 5 ORDER BY id")
    sub = ref[ref["id"] < 5]
    np.testing.assert_allclose(df3["total"].to_numpy(),
                               (sub["price"] * sub["qty"]).to_numpy())
    # unary minus and negative literals
    df4 = execute_sql(cat, "SELECT sum(-price) s FROM orders WHERE id = 1")
    assert abs(df4["s"].iloc[0] + ref[ref["id"] == 1]["price"].iloc[0]) < 1e-9


def test_tensor_vs_pandas_exec_equivalence(sql_table, monkeypatch):
    """Every query class produces identical results on the tensor engine
    and the legacy pandas path (LAKESOUL_SQL_PANDAS=1)."""
    cat, t = sql_table
    queries = [
        "SELECT region, count(*) n, sum(price) s, avg(qty) a FROM orders "
        "GROUP BY region ORDER BY region",
        "SELECT count(DISTINCT region) d FROM orders",
        "SELECT DISTINCT region FROM orders ORDER BY region",
        "SELECT id, price FROM orders WHERE qty > 5 ORDER BY price DESC, id LIMIT 7",
        "SELECT region, max(price) mx, min(qty) mn FROM orders "
        "GROUP BY region HAVING mx > 50 ORDER BY region",
        "SELECT sum(price * qty) v FROM orders",
    ]
    for sql in queries:
        monkeypatch.delenv("LAKESOUL_SQL_PANDAS", raising=False)
        a = execute_sql(cat, sql)
        monkeypatch.setenv("LAKESOUL_SQL_PANDAS", "1")
        b = execute_sql(cat, sql)
        monkeypatch.delenv("LAKESOUL_SQL_PANDAS", raising=False)
        assert list(a.columns) == list(b.columns), sql
        assert len(a) == len(b), sql
        for c in a.columns:
            av, bv = a[c].to_numpy(), b[c].to_numpy()
            if av.dtype.kind in "fc" or bv.dtype.kind in "fc":
                np.testing.assert_allclose(
                    av.astype(float), bv.astype(float), rtol=1e-9,
                    err_msg=f"{sql} :: {c}")
            else:
                np.testing.assert_array_equal(av, bv, err_msg=f"{sql} :: {c}")


def test_join_on_tensor_engine_matches_pandas(catalog, monkeypatch):
    execute_sql(catalog,
        "CREATE TABLE cust (cid BIGINT NOT NULL, name VARCHAR(16)) "
        "PRIMARY KEY (cid) HASH BUCKETS 2")
    execute_sql(catalog,
        "CREATE TABLE ords (oid BIGINT NOT NULL, cid BIGINT, amt DOUBLE) "
        "PRIMARY KEY (oid) HASH BUCKETS 2")
    execute_sql(catalog,
        "INSERT INTO cust VALUES (1, 'ann'), (2, 'bob'), (3, 'cyd')")
    execute_sql(catalog,
        "INSERT INTO ords VALUES (10, 1, 5.0), (11, 1, 7.0), (12, 2, 3.0), "
        "(13, NULL, 9.0)")
    sql = ("SELECT c.name, sum(o.amt) total FROM ords o "
           "JOIN cust c ON o.cid = c.cid GROUP BY c.name ORDER BY c.name")
    a = execute_sql(catalog, sql)
    monkeypatch.setenv("LAKESOUL_SQL_PANDAS", "1")
    b = execute_sql(catalog, sql)
    monkeypatch.delenv("LAKESOUL_SQL_PANDAS", raising=False)
    assert a["name"].tolist() == b["name"].tolist() == ["ann", "bob"]
    np.testing.assert_allclose(a["total"].to_numpy(), b["total"].to_numpy())
    assert a["total"].tolist() == [12.0, 3.0]
    # left join keeps the null-cid order with null name
    sql2 = ("SELECT o.oid, c.name FROM ords o LEFT JOIN cust c "
            "ON o.cid = c.cid ORDER BY o.oid")
    a2 = execute_sql(catalog, sql2)
    assert a2["oid"].tolist() == [10, 11, 12, 13]
    assert a2["name"].tolist()[:3] == ["ann", "ann", "bob"]
    assert a2["name"].iloc[3] is None or pd.isna(a2["name"].iloc[3])


def test_three_way_join(catalog, monkeypatch):
    """N-way join fold (the real TPC-H q3 shape: customer x orders x
    lineitem), tensor engine vs pandas oracle."""
    execute_sql(catalog,
        "CREATE TABLE cust3 (cid BIGINT NOT NULL, seg VARCHAR(8)) "
        "PRIMARY KEY (cid) HASH BUCKETS 2")
    execute_sql(catalog,
        "CREATE TABLE ord3 (oid BIGINT NOT NULL, cid BIGINT, odate BIGINT) "
        "PRIMARY KEY (oid) HASH BUCKETS 2")
    execute_sql(catalog,
        "CREATE TABLE li3 (lid BIGINT NOT NULL, oid BIGINT, amt DOUBLE) "
        "PRIMARY KEY (lid) HASH BUCKETS 2")
    execute_sql(catalog, "INSERT INTO cust3 VALUES (1,'a'),(2,'b'),(3,'a')")
    execute_sql(catalog,
        "INSERT INTO ord3 VALUES (10,1,100),(11,2,200),(12,3,300),(13,1,400)")
    execute_sql(catalog,
        "INSERT INTO li3 VALUES (7,10,5.0),(8,10,6.0),(9,11,2.0),(14,13,1.0),"
        "(15,12,9.0)")
    sql = ("SELECT c.seg, sum(l.amt) total, count(*) n "
           "FROM li3 l JOIN ord3 o ON l.oid = o.oid "
           "JOIN cust3 c ON o.cid = c.cid "
           "WHERE o.odate < 350 GROUP BY c.seg ORDER BY c.seg")
    a = execute_sql(catalog, sql)
    assert a["seg"].tolist() == ["a", "b"]
    # seg 'a': orders 10 (amt 5+6) + 12 (9.0) -> 20.0 ; 13 excluded (odate 400)
    np.testing.assert_allclose(a["total"].to_numpy(), [20.0, 2.0])
    assert a["n"].tolist() == [3, 1]
    monkeypatch.setenv("LAKESOUL_SQL_PANDAS", "1")
    b = execute_sql(catalog, sql)
    monkeypatch.delenv("LAKESOUL_SQL_PANDAS", raising=False)
    assert a["seg"].tolist() == b["seg"].tolist()
    np.testing.assert_allclose(a["total"].to_numpy(), b["total"].to_numpy())
    # bare-name disambiguation across three tables
    sql2 = ("SELECT lid, o.oid, o.cid FROM li3 JOIN ord3 o ON li3.oid = o.oid "
            "JOIN cust3 ON o.cid = cust3.cid ORDER BY lid")
    d2 = execute_sql(catalog, sql2)
    assert d2["lid"].tolist() == [7, 8, 9, 14, 15]
    assert d2["cid"].tolist() == [1, 1, 2, 1, 3]


def test_insert_null_values_and_decimal(catalog):
    """NULL in INSERT VALUES and INSERT..SELECT from a nullable decimal
    column (Decimal(str(None)) crash + literal() NULL)."""
    import decimal

    execute_sql(catalog,
        "CREATE TABLE nv (id BIGINT NOT NULL, amt DECIMAL(10,2), "
        "qty INT, note VARCHAR(16)) PRIMARY KEY (id) HASH BUCKETS 1")
    execute_sql(catalog,
        "INSERT INTO nv VALUES (1, 9.5, 10, 'a'), (2, NULL, NULL, NULL)")
    df = execute_sql(catalog, "SELECT id, amt, qty, note FROM nv ORDER BY id")
    assert df["amt"].tolist()[0] == decimal.Decimal("9.50")
    assert pd.isna(df["amt"].iloc[1])
    assert pd.isna(df["qty"].iloc[1])
    assert df["note"].iloc[1] is None or pd.isna(df["note"].iloc[1])
    # INSERT ... SELECT from the nullable decimal column must not abort
    execute_sql(catalog,
        "CREATE TABLE nv2 (id BIGINT NOT NULL, amt DECIMAL(10,2)) "
        "PRIMARY KEY (id) HASH BUCKETS 1")
    execute_sql(catalog, "INSERT INTO nv2 SELECT id, amt FROM nv")
    df2 = execute_sql(catalog, "SELECT id, amt FROM nv2 ORDER BY id")
    assert df2["amt"].tolist()[0] == decimal.Decimal("9.50")
    assert pd.isna(df2["amt"].iloc[1])


def test_alter_add_column_sql(sql_table):
    cat, t = sql_table
    execute_sql(cat, "ALTER TABLE orders ADD COLUMN note VARCHAR(20), score DOUBLE")
    t2 = cat.table("orders")
    assert "note" in t2.schema.names() and "score" in t2.schema.names()
    df = execute_sql(cat, "SELECT note, score FROM orders WHERE id = 1")
    assert df["note"].isna().iloc[0]


def test_compact_and_vacuum_sql(sql_table):
    cat, t = sql_table
    for it in range(3):
        t.upsert({"id": np.arange(50, dtype=np.int64),
                  "price": np.full(50, float(it)),
                  "qty": np.ones(50, dtype=np.int64), "region": ["east"] * 50})
    files_before = len(t.files())
    execute_sql(cat, "COMPACT TABLE orders")
    assert len(t.files()) < files_before
    r = execute_sql(cat, "VACUUM orders KEEP 1")
    assert r["files_removed"].iloc[0] >= 0
    df = execute_sql(cat, "SELECT count(*) n FROM orders")
    assert df["n"].iloc[0] == 1000


def test_show_partitions_and_history(sql_table):
    cat, t = sql_table
    df = execute_sql(cat, "SHOW PARTITIONS orders")
    assert df["partition"].tolist() == ["-5"]
    h = execute_sql(cat, "SHOW HISTORY orders")
    assert h["commit_op"].iloc[0] == "MergeCommit"
    assert (h["version"].diff().dropna() == 1).all()


def test_update_decimal_column(catalog):
    import decimal

    execute_sql(catalog,
        "CREATE TABLE updec (id BIGINT NOT NULL, amt DECIMAL(8,2)) "
        "PRIMARY KEY (id) HASH BUCKETS 1")
    execute_sql(catalog, "INSERT INTO updec VALUES (1, 1.00), (2, 2.00)")
    execute_sql(catalog, "UPDATE updec SET amt = 7.77 WHERE id = 2")
    df = execute_sql(catalog, "SELECT id, amt FROM updec ORDER BY id")
    assert df["amt"].tolist() == [decimal.Decimal("1.00"), decimal.Decimal("7.77")]


def test_show_create_table(catalog):
    execute_sql(catalog,
        "CREATE TABLE sct (id BIGINT NOT NULL, v DOUBLE, r VARCHAR(8)) "
        "PRIMARY KEY (id) HASH BUCKETS 4 PARTITION BY (r)")
    df = execute_sql(catalog, "SHOW CREATE TABLE sct")
    stmt = df["create_statement"].iloc[0]
    assert "id int64 NOT NULL" in stmt
    assert "PRIMARY KEY (id)" in stmt
    assert "HASH BUCKETS 4" in stmt
    assert "PARTITION BY (r)" in stmt
    # round trip: the emitted statement recreates an equivalent table
    execute_sql(catalog, stmt.replace("TABLE sct", "TABLE sct2"))
    t2 = catalog.table("sct2")
    assert t2.primary_keys == ["id"] and t2.hash_bucket_num == 4
    assert t2.range_keys == ["r"]


def test_create_table_nested_types_sql(catalog):
    """CREATE TABLE with canonical nested dtypes (list<string>,
    struct<...>, map<K,V>) — the names SHOW CREATE TABLE emits."""
    import numpy as np

    execute_sql(catalog,
        "CREATE TABLE nst (id BIGINT NOT NULL, tags list<string>, "
        "st struct<a:int64,b:string>, mp map<string,int64>) "
        "PRIMARY KEY (id) HASH BUCKETS 1")
    t = catalog.table("nst")
    assert t.schema.field("tags").dtype == "list<string>"
    assert t.schema.field("st").dtype == "struct<a:int64,b:string>"
    assert t.schema.field("mp").dtype == "map<string,int64>"
    t.upsert({"id": np.arange(2, dtype=np.int64),
              "tags": [["x"], None],
              "st": [{"a": 1, "b": "y"}, None],
              "mp": [{"k": 7}, None]})
    df = execute_sql(catalog, "SELECT id, st.a FROM nst ORDER BY id")
    assert df["a"].iloc[0] == 1
    # SHOW CREATE round-trips the nested types too
    stmt = execute_sql(catalog, "SHOW CREATE TABLE nst")["create_statement"].iloc[0]
    assert "list<string>" in stmt and "struct<a:int64,b:string>" in stmt
    execute_sql(catalog, stmt.replace("TABLE nst", "TABLE nst2"))
    assert catalog.table("nst2").schema.field("st").dtype == \
        "struct<a:int64,b:string>"
