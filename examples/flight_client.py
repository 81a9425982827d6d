#!/usr/bin/env python3
"""Arrow Flight gateway quickstart: start the server in-process, then act
as a remote engine — handshake, SQL over gRPC, streaming reads,
transactional writes.

    python examples/flight_client.py
"""

import json
import os
import sys
import tempfile

import numpy as np
import pyarrow as pa
import pyarrow.flight as fl

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

workdir = tempfile.mkdtemp(prefix="lakesoul_flight_")
os.environ["LAKESOUL_META_DB"] = os.path.join(workdir, "meta.db")
os.environ["LAKESOUL_WAREHOUSE"] = os.path.join(workdir, "warehouse")

from lakesoul_amd import Field, LakeSoulCatalog, Schema            # noqa: E402
from lakesoul_amd.service.flight_server import (                   # noqa: E402
    LakeSoulFlightServer, connect)

catalog = LakeSoulCatalog()
server = LakeSoulFlightServer("grpc://127.0.0.1:0", catalog=catalog)

client, opts = connect(f"grpc://127.0.0.1:{server.port}", "alice")

# create a table over gRPC
list(client.do_action(fl.Action("create_table", json.dumps({
    "table": "events",
    "schema": [{"name": "id", "type": "int64", "nullable": False},
               {"name": "v", "type": "float64"}],
    "primary_keys": ["id"], "hash_bucket_num": 2,
}).encode()), options=opts))

# transactional ingest (one commit per do_put stream)
tbl = pa.table({"id": pa.array(np.arange(1000), pa.int64()),
                "v": pa.array(np.random.default_rng(0).normal(size=1000))})
writer, meta_reader = client.do_put(
    fl.FlightDescriptor.for_command(json.dumps({"table": "events"}).encode()),
    tbl.schema, options=opts)
writer.write_table(tbl)
writer.done_writing()
print("ingest ack:", meta_reader.read().to_pybytes().decode())
writer.close()

# streaming read with filter pushdown
out = client.do_get(fl.Ticket(json.dumps({
    "table": "events", "columns": ["id"], "filters": [["id", "<", 5]],
}).encode()), options=opts).read_all()
print("filtered ids:", sorted(out.column("id").to_pylist()))

# SQL over gRPC
res = list(client.do_action(fl.Action("sql", json.dumps({
    "query": "SELECT count(*) AS n, avg(v) AS m FROM events"}).encode()),
    options=opts))
print("sql:", pa.ipc.open_stream(res[0].body.to_pybytes()).read_all().to_pydict())

server.shutdown()
print("flight OK —", workdir)
