"""RBAC object-store proxy.

Analog of the reference's ``rust/lakesoul-s3-proxy`` (pingora): an HTTP
proxy in front of the object store that parses the table path out of
S3-style requests (main.rs:668-710), verifies table-level RBAC against
the metadata catalog (verify_rbac, main.rs:204-258), counts Prometheus-
style metrics (main.rs:44-52), and forwards allowed requests to the
backing store. Here the backing store is the fs.py layer (mock:// in
tests, s3:// in production via pyarrow.fs), and auth tokens are the
gateway's HMAC tokens (service/server.py TokenService = jwt.rs analog).

    app = create_s3_proxy(catalog, backend_scheme="mock")
    GET/HEAD /{bucket}/{key...}   -> read through (RBAC-checked)
    PUT      /{bucket}/{key...}   -> write through (RBAC-checked)
"""

from __future__ import annotations

import os
from typing import Optional

try:
    from fastapi import FastAPI, HTTPException, Request, Response

    _HAVE_FASTAPI = True
except ImportError:  # pragma: no cover
    _HAVE_FASTAPI = False

from .server import TokenService


class ProxyMetrics:
    def __init__(self):
        self.requests = 0
        self.allowed = 0
        self.denied = 0
        self.bytes_in = 0
        self.bytes_out = 0

    def snapshot(self):
        return self.__dict__.copy()


def _table_path_of(catalog, full_path: str) -> Optional[str]:
    """Longest registered table_path that prefixes the object path
    (reference parses /table_path/part-....parquet the same way)."""
    best = None
    for ns in catalog.list_namespaces():
        for name in catalog.list_tables(ns):
            t = catalog.table(name, ns)
            tp = t.table_path
            if full_path.startswith(tp.rstrip("/") + "/") or full_path == tp:
                if best is None or len(tp) > len(best[0]):
                    best = (tp, t)
    return best


def create_s3_proxy(catalog, backend_scheme: str = "mock", secret: Optional[str] = None):
    if not _HAVE_FASTAPI:  # pragma: no cover
        raise ImportError("fastapi not available")
    from ..io.fs import FileSystem

    app = FastAPI(title="lakesoul_amd s3 proxy")
    tokens = TokenService(secret)
    metrics = ProxyMetrics()
    fs = FileSystem()

    def _auth_domain(request: Request) -> str:
        h = request.headers.get("authorization", "")
        if not h.startswith("Bearer "):
            raise HTTPException(401, "missing bearer token")
        try:
            return tokens.verify(h[len("Bearer "):]).get("domain", "public")
        except PermissionError as e:
            raise HTTPException(401, str(e))

    def _check(request: Request, bucket: str, key: str) -> str:
        metrics.requests += 1
        domain = _auth_domain(request)
        obj_path = f"{backend_scheme}://{bucket}/{key}"
        hit = _table_path_of(catalog, obj_path)
        if hit is not None:
            t = hit[1]
            if t.info.domain not in ("public", domain):
                metrics.denied += 1
                raise HTTPException(
                    403, f"domain {domain} cannot access table {t.info.table_name}"
                )
        metrics.allowed += 1
        return obj_path

    @app.get("/__metrics")
    def get_metrics():
        return metrics.snapshot()

    @app.get("/__metrics/prometheus")
    def get_metrics_prom():
        """Prometheus exposition format (reference s3-proxy exports
        IntCounters, main.rs:44-52)."""
        from fastapi.responses import PlainTextResponse

        lines = []
        for k, v in metrics.snapshot().items():
            name = f"lakesoul_s3proxy_{k}"
            lines.append(f"# TYPE {name} counter")
            lines.append(f"{name} {v}")
        return PlainTextResponse("\n".join(lines) + "\n")

    @app.get("/{bucket}/{key:path}")
    def get_object(bucket: str, key: str, request: Request):
        obj = _check(request, bucket, key)
        try:
            local = fs.localize(obj)
        except Exception:
            raise HTTPException(404, "no such object")
        data = open(local, "rb").read()
        metrics.bytes_out += len(data)
        return Response(content=data, media_type="application/octet-stream")

    @app.head("/{bucket}/{key:path}")
    def head_object(bucket: str, key: str, request: Request):
        obj = _check(request, bucket, key)
        try:
            local = fs.localize(obj)
        except Exception:
            raise HTTPException(404, "no such object")
        return Response(headers={"content-length": str(os.path.getsize(local))})

    @app.put("/{bucket}/{key:path}")
    async def put_object(bucket: str, key: str, request: Request):
        obj = _check(request, bucket, key)
        body = await request.body()
        metrics.bytes_in += len(body)
        import tempfile

        fd, tmp = tempfile.mkstemp()
        os.close(fd)
        with open(tmp, "wb") as f:
            f.write(body)
        fs.upload(tmp, obj)
        if os.path.exists(tmp):
            os.remove(tmp)
        return {"ok": True}

    app.state.metrics = metrics
    app.state.tokens = tokens
    return app
