"""Arrow Flight server tests (reference: rust/lakesoul-flight
flight_sql_service.rs — handshake auth, do_get reads, do_put ingest,
actions, RBAC)."""

import json

import numpy as np
import pyarrow as pa
import pytest

fl = pytest.importorskip("pyarrow.flight")

from lakesoul_amd.io.schema import Field, Schema
from lakesoul_amd.service.flight_server import LakeSoulFlightServer, connect


@pytest.fixture
def flight(catalog):
    t = catalog.create_table(
        "ft",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    t.upsert({"id": np.arange(100, dtype=np.int64),
              "v": np.arange(100, dtype=np.float64)})
    srv = LakeSoulFlightServer("grpc://127.0.0.1:0", catalog=catalog, secret="s3")
    yield srv, catalog
    srv.shutdown()


def _client(srv, user="alice", domain="public"):
    return connect(f"grpc://127.0.0.1:{srv.port}", user, domain)


def test_handshake_and_do_get(flight):
    srv, catalog = flight
    client, opts = _client(srv)
    ticket = fl.Ticket(json.dumps({"table": "ft"}).encode())
    t = client.do_get(ticket, options=opts).read_all()
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == 100 and df["v"].iloc[42] == 42.0


def test_unauthenticated_rejected(flight):
    srv, _ = flight
    client = fl.connect(f"grpc://127.0.0.1:{srv.port}")
    with pytest.raises(fl.FlightUnauthenticatedError):
        client.do_get(fl.Ticket(json.dumps({"table": "ft"}).encode())).read_all()
    # garbage token too
    bad = fl.FlightCallOptions(headers=[(b"authorization", b"Bearer junk")])
    with pytest.raises(fl.FlightUnauthenticatedError):
        client.do_get(fl.Ticket(json.dumps({"table": "ft"}).encode()),
                      options=bad).read_all()


def test_do_get_filters_columns_version(flight):
    srv, _ = flight
    client, opts = _client(srv)
    ticket = fl.Ticket(json.dumps({
        "table": "ft", "columns": ["id"], "filters": [["id", "<", 5]],
    }).encode())
    t = client.do_get(ticket, options=opts).read_all()
    assert t.column_names == ["id"]
    assert sorted(t.column("id").to_pylist()) == [0, 1, 2, 3, 4]


def test_do_put_upsert(flight):
    srv, catalog = flight
    client, opts = _client(srv)
    desc = fl.FlightDescriptor.for_command(json.dumps({"table": "ft"}).encode())
    tbl = pa.table({"id": pa.array([1, 2], pa.int64()),
                    "v": pa.array([-1.0, -2.0], pa.float64())})
    writer, meta_reader = client.do_put(desc, tbl.schema, options=opts)
    writer.write_table(tbl)
    writer.done_writing()
    ack = json.loads(meta_reader.read().to_pybytes())
    assert ack["rows"] == 2
    writer.close()
    df = catalog.table("ft").to_pandas()
    assert df[df.id == 1]["v"].iloc[0] == -1.0
    assert srv.metrics.total_rows == 2


def test_flight_info_and_list(flight):
    srv, _ = flight
    client, opts = _client(srv)
    info = client.get_flight_info(
        fl.FlightDescriptor.for_path("default", "ft"), options=opts)
    assert info.schema.names == ["id", "v"]
    flights = list(client.list_flights(options=opts))
    assert any(
        json.loads(f.endpoints[0].ticket.ticket.decode())["table"] == "ft"
        for f in flights
    )


def test_actions_create_compact_sql_metrics(flight):
    srv, catalog = flight
    client, opts = _client(srv)
    # create_table
    res = list(client.do_action(fl.Action("create_table", json.dumps({
        "table": "t2",
        "schema": [{"name": "k", "type": "int64", "nullable": False},
                   {"name": "x", "type": "float64"}],
        "primary_keys": ["k"], "hash_bucket_num": 1,
    }).encode()), options=opts))
    assert json.loads(res[0].body.to_pybytes())["table_id"]
    assert catalog.table_exists("t2")
    # sql action returns arrow IPC
    res = list(client.do_action(fl.Action("sql", json.dumps(
        {"query": "SELECT count(*) AS n FROM ft"}).encode()), options=opts))
    buf = res[0].body.to_pybytes()
    t = pa.ipc.open_stream(buf).read_all()
    assert t.column("n").to_pylist() == [100]
    # compaction + metrics
    list(client.do_action(fl.Action("compaction", json.dumps(
        {"table": "ft"}).encode()), options=opts))
    res = list(client.do_action(fl.Action("metrics", b""), options=opts))
    m = json.loads(res[0].body.to_pybytes())
    assert m["requests"] >= 0


def test_rbac_domain_enforced(flight, catalog):
    srv, _ = flight
    t = catalog.create_table(
        "priv",
        Schema([Field("id", "int64", False)]),
        primary_keys=["id"],
        hash_bucket_num=1,
        properties={"domain": "teamA"},
    )
    if t.info.domain == "public":
        pytest.skip("catalog does not store domains via properties")
    client, opts = _client(srv, domain="teamB")
    with pytest.raises((fl.FlightUnauthorizedError, fl.FlightServerError)):
        client.do_get(fl.Ticket(json.dumps({"table": "priv"}).encode()),
                      options=opts).read_all()
    client2, opts2 = _client(srv, domain="teamA")
    t2 = client2.do_get(fl.Ticket(json.dumps({"table": "priv"}).encode()),
                        options=opts2).read_all()
    assert t2.num_rows == 0


def test_do_get_incremental(flight):
    """Streaming readers poll incremental tickets (reference: Flink
    LakeSoulSource dynamic splits over new commits)."""
    srv, catalog = flight
    t = catalog.table("ft")
    import numpy as np

    t.upsert({"id": np.array([1000, 1001], dtype=np.int64),
              "v": np.array([1.0, 2.0])})
    client, opts = _client(srv)
    ticket = fl.Ticket(json.dumps({"table": "ft", "incremental": [0, 10**9]}).encode())
    inc = client.do_get(ticket, options=opts).read_all()
    ids = set(inc.column("id").to_pylist())
    assert {1000, 1001} <= ids
    assert len(ids) < 102  # not the full table


def test_flightsql_statement_query_wire(flight):
    """The standard Flight SQL GetFlightInfo/DoGet flow with wire-exact
    Any-packed CommandStatementQuery / TicketStatementQuery envelopes
    (reference flight_sql_service.rs:218)."""
    from lakesoul_amd.service import flightsql as fsql

    srv, catalog = flight
    client, opts = _client(srv)
    desc = fl.FlightDescriptor.for_command(
        fsql.cmd_statement_query("SELECT id, v FROM ft WHERE id < 5 ORDER BY id"))
    info = client.get_flight_info(desc, options=opts)
    assert info.total_records == 5
    ticket = info.endpoints[0].ticket
    # the ticket must be an Any-packed TicketStatementQuery
    name, payload = fsql.unpack_any(ticket.ticket)
    assert name == "TicketStatementQuery"
    assert fsql.parse_bytes_field(payload, 1)
    t = client.do_get(ticket, options=opts).read_all()
    assert t.column("id").to_pylist() == [0, 1, 2, 3, 4]
    # a second do_get with the same handle fails (handle consumed)
    with pytest.raises(fl.FlightError):
        client.do_get(ticket, options=opts).read_all()


def test_flightsql_catalog_metadata_commands(flight):
    from lakesoul_amd.service import flightsql as fsql

    srv, catalog = flight
    client, opts = _client(srv)
    for cmd, col, expect in [
        (fsql.cmd_get_catalogs(), "catalog_name", ["lakesoul"]),
        (fsql.cmd_get_db_schemas(), "db_schema_name", None),
        (fsql.cmd_get_tables(), "table_name", None),
    ]:
        info = client.get_flight_info(fl.FlightDescriptor.for_command(cmd),
                                      options=opts)
        t = client.do_get(info.endpoints[0].ticket, options=opts).read_all()
        assert col in t.schema.names
        if expect is not None:
            assert t.column(col).to_pylist() == expect
    # tables listing includes ft
    info = client.get_flight_info(
        fl.FlightDescriptor.for_command(fsql.cmd_get_tables()), options=opts)
    t = client.do_get(info.endpoints[0].ticket, options=opts).read_all()
    assert "ft" in t.column("table_name").to_pylist()


def test_flightsql_statement_update(flight):
    from lakesoul_amd.service import flightsql as fsql

    srv, catalog = flight
    client, opts = _client(srv)
    desc = fl.FlightDescriptor.for_command(
        fsql.cmd_statement_update("INSERT INTO ft VALUES (1000, 42.0)"))
    schema = pa.schema([])
    writer, reader = client.do_put(desc, schema, options=opts)
    writer.done_writing()
    buf = reader.read()
    assert fsql.parse_do_put_update_result(bytes(memoryview(buf))) == 1
    writer.close()
    # row landed
    ticket = fl.Ticket(json.dumps({"table": "ft"}).encode())
    t = client.do_get(ticket, options=opts).read_all()
    assert 1000 in t.column("id").to_pylist()
