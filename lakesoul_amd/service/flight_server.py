"""Arrow Flight server — the gRPC analog of the reference's
``lakesoul-flight`` crate (``flight_sql_service.rs:218-1085``: handshake/JWT
auth, get_flight_info, do_get streaming reads, do_put transactional ingest,
``StreamWriteMetrics``).

The reference implements the Flight **SQL** extension over tonic; here the
plain Flight RPCs carry the same operations (pyarrow.flight is the gRPC
stack in this image):

- handshake(username[,domain])      -> HMAC token (jwt.rs analog)
- list_flights / get_flight_info    -> table discovery + schema
- do_get(ticket JSON)               -> MOR scan stream (columns/filters/
                                       version/shard pushdown)
- do_put(descriptor JSON)           -> upsert/append ingest, committed
                                       atomically per stream (2PC commit
                                       protocol underneath)
- do_action                         -> create_table / compaction / metrics
                                       / sql (console queries over gRPC)

RBAC: the table's ``domain`` must match the token's domain
(rbac.rs:19-50 ``verify_permission_by_table_name`` analog).
"""

from __future__ import annotations

import json
import threading
from typing import Optional

try:
    import pyarrow as pa
    import pyarrow.flight as fl

    _HAVE_FLIGHT = True
except ImportError:  # pragma: no cover
    _HAVE_FLIGHT = False

from .server import StreamWriteMetrics, TokenService


class _AuthMiddlewareFactory(fl.ServerMiddlewareFactory if _HAVE_FLIGHT else object):
    """Bearer-token check on every call except Handshake (which issues
    tokens). The verified claims ride on the middleware instance."""

    def __init__(self, tokens: TokenService):
        self.tokens = tokens

    # token issuance itself flows through do_action("handshake"), so
    # DO_ACTION passes here and is checked per-action in do_action()
    _OPEN_METHODS = ("HANDSHAKE", "DO_ACTION", "LIST_ACTIONS")

    def start_call(self, info, headers):
        auth = None
        for k, v in headers.items():
            if k.lower() == "authorization":
                auth = v[0] if isinstance(v, (list, tuple)) else v
        open_call = any(
            info.method == getattr(fl.FlightMethod, m) for m in self._OPEN_METHODS
        )
        if not auth or not auth.startswith("Bearer "):
            if open_call:
                return None
            raise fl.FlightUnauthenticatedError("missing bearer token")
        try:
            claims = self.tokens.verify(auth[len("Bearer "):])
        except PermissionError as e:
            if open_call:
                return None
            raise fl.FlightUnauthenticatedError(str(e))
        return _AuthMiddleware(claims)


class _AuthMiddleware(fl.ServerMiddleware if _HAVE_FLIGHT else object):
    def __init__(self, claims: dict):
        self.claims = claims


class LakeSoulFlightServer(fl.FlightServerBase if _HAVE_FLIGHT else object):
    def __init__(self, location: str = "grpc://127.0.0.1:0", catalog=None,
                 secret: Optional[str] = None):
        if not _HAVE_FLIGHT:  # pragma: no cover
            raise ImportError("pyarrow.flight not available")
        if catalog is None:
            from ..tables.catalog import LakeSoulCatalog

            catalog = LakeSoulCatalog()
        self.catalog = catalog
        self.tokens = TokenService(secret)
        self.metrics = StreamWriteMetrics()
        self._lock = threading.Lock()
        self._sql_results: dict = {}  # statement handle -> result table
        super().__init__(
            location,
            middleware={"auth": _AuthMiddlewareFactory(self.tokens)},
        )

    # -- auth ----------------------------------------------------------- #

    def _claims(self, context) -> dict:
        mw = context.get_middleware("auth")
        return mw.claims if mw is not None else {}

    def _check_domain(self, table, claims: dict, write: bool = False):
        domain = claims.get("domain", "public")
        if table.info.domain not in ("public", domain):
            raise fl.FlightUnauthorizedError(
                f"domain {domain} cannot access table domain {table.info.domain}"
            )

    # -- discovery ------------------------------------------------------ #

    def _table(self, name: str, namespace: str, claims: dict, write=False):
        t = self.catalog.table(name, namespace)
        self._check_domain(t, claims, write)
        return t

    def list_flights(self, context, criteria):
        claims = self._claims(context)
        ns_list = self.catalog.list_namespaces() or ["default"]
        for ns in ns_list:
            for name in self.catalog.list_tables(ns):
                try:
                    t = self._table(name, ns, claims)
                except fl.FlightUnauthorizedError:
                    continue
                yield self._flight_info(t, {"table": name, "namespace": ns})

    def _flight_info(self, t, ticket_dict: dict):
        from ..io.schema import schema_to_arrow

        ticket = fl.Ticket(json.dumps(ticket_dict).encode())
        desc = fl.FlightDescriptor.for_path(
            ticket_dict["namespace"], ticket_dict["table"]
        )
        ep = fl.FlightEndpoint(ticket, [])
        n_rows = -1
        return fl.FlightInfo(schema_to_arrow(t.schema), desc, [ep], n_rows, -1)

    # -- Flight SQL protocol (wire-level; service/flightsql.py) -------- #

    def _try_flightsql(self, cmd: bytes):
        """Return (short_type, payload) if cmd is a flight-sql Any."""
        try:
            from . import flightsql as fsql

            name, payload = fsql.unpack_any(cmd)
            return (name, payload) if name else (None, b"")
        except Exception:
            return (None, b"")

    def _flightsql_info(self, context, name: str, payload: bytes, descriptor):
        """GetFlightInfo for flight-sql commands: run/prepare the result,
        stash it under a statement handle, return endpoints whose ticket
        is an Any-packed TicketStatementQuery (flight_sql_service.rs:218
        get_flight_info_statement analog)."""
        import uuid

        from . import flightsql as fsql

        if name == "CommandStatementQuery":
            from ..sql import execute_sql

            query = fsql.parse_string_field(payload, 1)
            df = execute_sql(self.catalog, query)
            tbl = pa.Table.from_pandas(df, preserve_index=False)
        elif name == "CommandGetCatalogs":
            tbl = pa.table({"catalog_name": pa.array(
                ["lakesoul"], pa.string())})
        elif name == "CommandGetDbSchemas":
            ns = self.catalog.list_namespaces() or ["default"]
            tbl = pa.table({
                "catalog_name": pa.array(["lakesoul"] * len(ns), pa.string()),
                "db_schema_name": pa.array(ns, pa.string()),
            })
        elif name == "CommandGetTables":
            rows = {"catalog_name": [], "db_schema_name": [],
                    "table_name": [], "table_type": []}
            for ns in self.catalog.list_namespaces() or ["default"]:
                for tname in self.catalog.list_tables(ns):
                    rows["catalog_name"].append("lakesoul")
                    rows["db_schema_name"].append(ns)
                    rows["table_name"].append(tname)
                    rows["table_type"].append("TABLE")
            tbl = pa.table({k: pa.array(v, pa.string())
                            for k, v in rows.items()})
        else:
            raise fl.FlightServerError(f"unsupported flight-sql command {name}")
        handle = uuid.uuid4().bytes
        with self._lock:
            self._sql_results[handle] = tbl
        ticket = fl.Ticket(fsql.ticket_statement_query(handle))
        ep = fl.FlightEndpoint(ticket, [])
        return fl.FlightInfo(tbl.schema, descriptor, [ep], tbl.num_rows, -1)

    def get_flight_info(self, context, descriptor):
        claims = self._claims(context)
        if descriptor.descriptor_type == fl.DescriptorType.CMD:
            name, payload = self._try_flightsql(descriptor.command)
            if name:
                return self._flightsql_info(context, name, payload, descriptor)
            d = json.loads(descriptor.command.decode())
        else:
            path = [p.decode() if isinstance(p, bytes) else p for p in descriptor.path]
            d = {"namespace": path[0], "table": path[1]} if len(path) > 1 else {
                "namespace": "default", "table": path[0]}
        t = self._table(d["table"], d.get("namespace", "default"), claims)
        return self._flight_info(t, d)

    def get_schema(self, context, descriptor):
        info = self.get_flight_info(context, descriptor)
        return fl.SchemaResult(info.schema)

    # -- reads ---------------------------------------------------------- #

    def do_get(self, context, ticket):
        claims = self._claims(context)
        name, payload = self._try_flightsql(ticket.ticket)
        if name == "TicketStatementQuery":
            from . import flightsql as fsql

            handle = fsql.parse_bytes_field(payload, 1)
            with self._lock:
                tbl = self._sql_results.pop(handle, None)
            if tbl is None:
                raise fl.FlightServerError("unknown statement handle")
            self.metrics.requests += 1
            return fl.RecordBatchStream(tbl)
        d = json.loads(ticket.ticket.decode())
        t = self._table(d["table"], d.get("namespace", "default"), claims)
        scan = t.scan(
            columns=d.get("columns"),
            filters=d.get("filters"),
            version=d.get("version"),
            partitions=d.get("partitions"),
            batch_size=d.get("batch_size"),
            incremental=tuple(d["incremental"]) if d.get("incremental") else None,
        )
        if d.get("world_size"):
            scan = scan.shard(int(d.get("rank", 0)), int(d["world_size"]))
        arrow_tbl = scan.to_arrow()
        self.metrics.requests += 1
        return fl.RecordBatchStream(arrow_tbl)

    # -- writes (transactional per stream) ------------------------------ #

    def do_put(self, context, descriptor, reader, writer):
        claims = self._claims(context)
        if descriptor.descriptor_type == fl.DescriptorType.CMD:
            name, payload = self._try_flightsql(descriptor.command)
            if name == "CommandStatementUpdate":
                # flight-sql UPDATE/INSERT path: run the statement, reply
                # with DoPutUpdateResult{record_count}
                from . import flightsql as fsql
                from ..sql import execute_sql

                query = fsql.parse_string_field(payload, 1)
                df = execute_sql(self.catalog, query)
                count = 0
                for c in ("rows_inserted", "rows_updated", "rows_deleted"):
                    if c in getattr(df, "columns", []):
                        count = int(df[c].iloc[0])
                        break
                writer.write(pa.py_buffer(fsql.do_put_update_result(count)))
                return
            d = json.loads(descriptor.command.decode())
        else:
            path = [p.decode() if isinstance(p, bytes) else p for p in descriptor.path]
            d = {"namespace": path[0], "table": path[1]} if len(path) > 1 else {
                "namespace": "default", "table": path[0]}
        t = self._table(d["table"], d.get("namespace", "default"), claims, write=True)
        batches = [b.data for b in reader]
        if not batches:
            writer.write(pa.py_buffer(json.dumps({"rows": 0}).encode()))
            return
        tbl = pa.Table.from_batches(batches)
        mode = d.get("mode", "upsert" if t.primary_keys else "append")
        with self._lock:
            self.metrics.active_streams += 1
        try:
            # one metadata commit for the whole stream = the reference's
            # transactional do_put (flight_sql_service.rs do_put ingest)
            if mode == "upsert":
                t.upsert(tbl)
            else:
                t.write(tbl)
        finally:
            with self._lock:
                self.metrics.active_streams -= 1
                self.metrics.total_rows += tbl.num_rows
                self.metrics.total_bytes += tbl.nbytes
        writer.write(pa.py_buffer(json.dumps({"rows": tbl.num_rows}).encode()))

    # -- actions --------------------------------------------------------- #

    def list_actions(self, context):
        return [
            ("handshake", "issue a token: {username, domain?}"),
            ("create_table", "{table, namespace?, schema, primary_keys?, "
                             "hash_bucket_num?, range_partitions?}"),
            ("compaction", "{table, namespace?}"),
            ("metrics", "stream-write counters"),
            ("sql", "run a SQL statement, returns Arrow IPC"),
        ]

    def do_action(self, context, action):
        body = action.body.to_pybytes() if action.body is not None else b""
        d = json.loads(body.decode()) if body else {}
        if action.type == "handshake":
            user = d.get("username", "")
            if not user:
                raise fl.FlightServerError("username required")
            token = self.tokens.issue(user, d.get("domain", "public"))
            return [json.dumps({"token": token}).encode()]
        claims = self._claims(context)
        if not claims:
            raise fl.FlightUnauthenticatedError("missing bearer token")
        if action.type == "metrics":
            return [json.dumps(self.metrics.snapshot()).encode()]
        if action.type == "create_table":
            from ..io.schema import Schema, normalize_schema

            sch = normalize_schema([(f["name"], f["type"], f.get("nullable", True))
                                    for f in d["schema"]])
            t = self.catalog.create_table(
                d["table"], sch,
                primary_keys=d.get("primary_keys") or [],
                hash_bucket_num=d.get("hash_bucket_num", 4),
                range_partitions=d.get("range_partitions") or [],
                namespace=d.get("namespace", "default"),
            )
            return [json.dumps({"table_id": t.table_id}).encode()]
        if action.type == "compaction":
            t = self._table(d["table"], d.get("namespace", "default"), claims, True)
            t.compaction()
            return [b"{}"]
        if action.type == "sql":
            import io as _io

            from ..sql import execute_sql

            df = execute_sql(self.catalog, d["query"])
            tbl = pa.Table.from_pandas(df, preserve_index=False)
            sink = _io.BytesIO()
            with pa.ipc.new_stream(sink, tbl.schema) as w:
                w.write_table(tbl)
            return [sink.getvalue()]
        raise fl.FlightServerError(f"unknown action {action.type}")


def connect(uri: str, username: str, domain: str = "public"):
    """Client helper: handshake and return (client, call_options)."""
    client = fl.connect(uri)
    res = list(client.do_action(fl.Action(
        "handshake", json.dumps({"username": username, "domain": domain}).encode())))
    token = json.loads(res[0].body.to_pybytes())["token"]
    opts = fl.FlightCallOptions(headers=[(b"authorization", f"Bearer {token}".encode())])
    return client, opts


def main():  # pragma: no cover
    import argparse

    ap = argparse.ArgumentParser()
    ap.add_argument("--host", default="0.0.0.0")
    ap.add_argument("--port", type=int, default=50051)
    args = ap.parse_args()
    srv = LakeSoulFlightServer(f"grpc://{args.host}:{args.port}")
    print(f"flight server on {args.host}:{srv.port}")
    srv.serve()


if __name__ == "__main__":  # pragma: no cover
    main()
