"""Serving gateway: Arrow-IPC over HTTP.

MI355X-native analog of the reference's Arrow Flight SQL server
(``rust/lakesoul-flight/src/flight_sql_service.rs:218-1085``): token
handshake, table listing/metadata, streaming reads (Arrow IPC), and
transactional ingest (two-phase commit on the metadata layer), plus the
throughput metrics the reference exposes (StreamWriteMetrics analog).

FastAPI + uvicorn replace tonic/gRPC (same control-plane role; the data
plane is Arrow IPC bytes either way). Run:

    python -m lakesoul_amd.service.server --host 0.0.0.0 --port 8850
"""

from __future__ import annotations

import base64
import hashlib
import hmac
import io
import json
import os
import time
from typing import Optional

try:
    from fastapi import Depends, FastAPI, HTTPException, Request, Response
    from fastapi.responses import StreamingResponse

    _HAVE_FASTAPI = True
except ImportError:  # pragma: no cover
    _HAVE_FASTAPI = False


class TokenService:
    """HMAC tokens (JWT-shaped) — analog of the reference's jwt.rs."""

    def __init__(self, secret: Optional[str] = None, ttl_s: int = 3600):
        self.secret = (secret or os.environ.get("LAKESOUL_TOKEN_SECRET", "lakesoul")).encode()
        self.ttl_s = ttl_s

    def issue(self, user: str, domain: str = "public") -> str:
        payload = json.dumps(
            {"sub": user, "domain": domain, "exp": int(time.time()) + self.ttl_s}
        ).encode()
        b = base64.urlsafe_b64encode(payload).rstrip(b"=")
        sig = base64.urlsafe_b64encode(
            hmac.new(self.secret, b, hashlib.sha256).digest()
        ).rstrip(b"=")
        return (b + b"." + sig).decode()

    def verify(self, token: str) -> dict:
        try:
            b, sig = token.encode().split(b".")
            expect = base64.urlsafe_b64encode(
                hmac.new(self.secret, b, hashlib.sha256).digest()
            ).rstrip(b"=")
            if not hmac.compare_digest(sig, expect):
                raise ValueError("bad signature")
            payload = json.loads(base64.urlsafe_b64decode(b + b"=" * (-len(b) % 4)))
            if payload.get("exp", 0) < time.time():
                raise ValueError("expired")
            return payload
        except Exception as e:
            raise PermissionError(f"invalid token: {e}")


class StreamWriteMetrics:
    """Counters analog of flight_sql_service.rs:80-216."""

    def __init__(self):
        self.active_streams = 0
        self.total_rows = 0
        self.total_bytes = 0
        self.requests = 0

    def snapshot(self) -> dict:
        return {
            "active_streams": self.active_streams,
            "total_rows": self.total_rows,
            "total_bytes": self.total_bytes,
            "requests": self.requests,
        }


def create_app(catalog=None, secret: Optional[str] = None):
    if not _HAVE_FASTAPI:  # pragma: no cover
        raise ImportError("fastapi not available")
    import pyarrow as pa
    import pyarrow.ipc as ipc

    if catalog is None:
        from ..tables.catalog import LakeSoulCatalog

        catalog = LakeSoulCatalog()

    app = FastAPI(title="lakesoul_amd gateway")
    tokens = TokenService(secret)
    metrics = StreamWriteMetrics()

    def auth(request: Request) -> dict:
        metrics.requests += 1
        h = request.headers.get("authorization", "")
        if not h.startswith("Bearer "):
            raise HTTPException(401, "missing bearer token")
        try:
            return tokens.verify(h[len("Bearer "):])
        except PermissionError as e:
            raise HTTPException(401, str(e))

    @app.post("/handshake")
    def handshake(body: dict):
        # reference: username/password handshake issuing a JWT
        user = body.get("username", "")
        if not user:
            raise HTTPException(400, "username required")
        return {"token": tokens.issue(user, body.get("domain", "public"))}

    @app.get("/namespaces")
    def namespaces(claims: dict = Depends(auth)):
        return {"namespaces": catalog.list_namespaces()}

    @app.get("/tables")
    def tables(namespace: str = "default", claims: dict = Depends(auth)):
        return {"tables": catalog.list_tables(namespace)}

    @app.get("/table/{name}/schema")
    def table_schema(name: str, namespace: str = "default", claims: dict = Depends(auth)):
        t = catalog.table(name, namespace)
        _check_domain(t, claims)
        return json.loads(t.info.table_schema) | {
            "primary_keys": t.primary_keys,
            "range_partitions": t.range_keys,
            "hash_bucket_num": t.hash_bucket_num,
        }

    @app.get("/table/{name}/scan")
    def table_scan(
        name: str,
        namespace: str = "default",
        columns: Optional[str] = None,
        filters: Optional[str] = None,
        version: Optional[int] = None,
        claims: dict = Depends(auth),
    ):
        t = catalog.table(name, namespace)
        _check_domain(t, claims)
        scan = t.scan(
            columns=columns.split(",") if columns else None,
            filters=filters,
            version=version,
        )

        def gen():
            metrics.active_streams += 1
            try:
                sink = io.BytesIO()
                writer = None
                for batch in scan.iter_batches():
                    tbl = batch.to_arrow()
                    for rb in tbl.to_batches():
                        if writer is None:
                            writer = ipc.new_stream(sink, rb.schema)
                        writer.write_batch(rb)
                        metrics.total_rows += rb.num_rows
                        data = sink.getvalue()
                        sink.seek(0)
                        sink.truncate()
                        metrics.total_bytes += len(data)
                        yield data
                if writer is not None:
                    writer.close()
                    yield sink.getvalue()
            finally:
                metrics.active_streams -= 1

        return StreamingResponse(gen(), media_type="application/vnd.apache.arrow.stream")

    @app.post("/table/{name}/write")
    async def table_write(name: str, request: Request, namespace: str = "default",
                          claims: dict = Depends(auth)):
        t = catalog.table(name, namespace)
        _check_domain(t, claims, write=True)
        body = await request.body()
        reader = ipc.open_stream(body)
        tbl = reader.read_all()
        t.write(tbl)
        metrics.total_rows += tbl.num_rows
        metrics.total_bytes += len(body)
        return {"rows": tbl.num_rows}

    @app.get("/table/{name}/splits")
    def table_splits(name: str, namespace: str = "default",
                     claims: dict = Depends(auth)):
        """Scan-plan units (SplitDesc analog, transfusion.rs:316): one
        entry per (partition, hash bucket) with the ordered file list —
        what an external engine needs to plan a distributed read."""
        t = catalog.table(name, namespace)
        _check_domain(t, claims)
        return {"splits": [
            {"partition_desc": u.partition_desc, "hash_bucket": u.bucket_id,
             "file_paths": u.files, "primary_keys": t.primary_keys}
            for u in t.scan().plan()]}

    @app.post("/table/{name}/compaction")
    def table_compaction(name: str, namespace: str = "default", claims: dict = Depends(auth)):
        t = catalog.table(name, namespace)
        _check_domain(t, claims, write=True)
        t.compaction()
        return {"ok": True}

    @app.post("/sql")
    async def run_sql(request: Request, claims: dict = Depends(auth)):
        """Run a SQL statement; returns rows as JSON (the Flight gateway
        streams Arrow IPC for bulk — this endpoint is the console/BI
        convenience surface)."""
        from fastapi.responses import JSONResponse

        from ..sql import SqlError, execute_sql

        body = await request.json()
        query = body.get("query", "")
        if not query:
            raise HTTPException(400, "query required")
        try:
            df = execute_sql(catalog, query)
        except SqlError as e:
            raise HTTPException(400, str(e))
        return JSONResponse({
            "columns": list(df.columns),
            "rows": json.loads(df.to_json(orient="values")),
        })

    @app.get("/metrics")
    def get_metrics(claims: dict = Depends(auth)):
        return metrics.snapshot()

    @app.get("/metrics/memory")
    def get_metrics_memory(claims: dict = Depends(auth)):
        """Host RSS + HBM allocator snapshot (jemalloc-prof analog,
        reference mem.rs:1-33)."""
        from ..utils import memprof

        return memprof.snapshot()

    @app.get("/metrics/prometheus")
    def get_metrics_prom(claims: dict = Depends(auth)):
        """Prometheus exposition format (reference: the s3-proxy and
        flight server export prometheus counters, main.rs:44-52)."""
        from fastapi.responses import PlainTextResponse

        snap = metrics.snapshot()
        lines = []
        for k, v in snap.items():
            name = f"lakesoul_{k}"
            kind = "gauge" if k == "active_streams" else "counter"
            lines.append(f"# TYPE {name} {kind}")
            lines.append(f"{name} {v}")
        return PlainTextResponse("\n".join(lines) + "\n")

    def _check_domain(t, claims: dict, write: bool = False):
        """RBAC: table domain must match the token's domain (analog of
        rbac.rs:19-50 verify_permission_by_table_name)."""
        domain = claims.get("domain", "public")
        if t.info.domain not in ("public", domain):
            raise HTTPException(403, f"domain {domain} cannot access table domain {t.info.domain}")

    app.state.metrics = metrics
    app.state.tokens = tokens
    return app


def main():  # pragma: no cover
    import argparse

    import uvicorn

    p = argparse.ArgumentParser()
    p.add_argument("--host", default="127.0.0.1")
    p.add_argument("--port", type=int, default=8850)
    args = p.parse_args()
    uvicorn.run(create_app(), host=args.host, port=args.port)


if __name__ == "__main__":  # pragma: no cover
    main()
