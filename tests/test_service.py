"""Serving gateway tests (fastapi TestClient, no network) + CLI tests."""

import io

import numpy as np
import pytest

from lakesoul_amd.io.schema import Field, Schema


@pytest.fixture()
def app_client(catalog):
    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from lakesoul_amd.service.server import create_app

    t = catalog.create_table(
        "served",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 1000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.arange(n, dtype=np.float64)})
    app = create_app(catalog, secret="testsecret")
    return TestClient(app)


def _token(client, user="alice", domain="public"):
    r = client.post("/handshake", json={"username": user, "domain": domain})
    assert r.status_code == 200
    return {"Authorization": f"Bearer {r.json()['token']}"}


def test_handshake_and_auth(app_client):
    assert app_client.get("/tables").status_code == 401
    hdr = _token(app_client)
    r = app_client.get("/tables", headers=hdr)
    assert r.status_code == 200
    assert "served" in r.json()["tables"]


def test_schema_endpoint(app_client):
    hdr = _token(app_client)
    r = app_client.get("/table/served/schema", headers=hdr)
    assert r.status_code == 200
    assert r.json()["primary_keys"] == ["id"]


def test_scan_stream_arrow_ipc(app_client):
    import pyarrow.ipc as ipc

    hdr = _token(app_client)
    r = app_client.get("/table/served/scan", headers=hdr)
    assert r.status_code == 200
    tbl = ipc.open_stream(io.BytesIO(r.content)).read_all()
    assert tbl.num_rows == 1000
    r = app_client.get(
        "/table/served/scan", headers=hdr, params={"filters": "lt(id, 10)"}
    )
    tbl = ipc.open_stream(io.BytesIO(r.content)).read_all()
    assert tbl.num_rows == 10


def test_write_ingest_two_phase(app_client):
    import pyarrow as pa
    import pyarrow.ipc as ipc

    hdr = _token(app_client)
    new = pa.table({"id": pa.array([5000, 5001], pa.int64()), "v": pa.array([1.0, 2.0])})
    sink = io.BytesIO()
    with ipc.new_stream(sink, new.schema) as w:
        w.write_table(new)
    r = app_client.post("/table/served/write", headers=hdr, content=sink.getvalue())
    assert r.status_code == 200 and r.json()["rows"] == 2
    r = app_client.get("/table/served/scan", headers=hdr)
    tbl = ipc.open_stream(io.BytesIO(r.content)).read_all()
    assert tbl.num_rows == 1002
    m = app_client.get("/metrics", headers=hdr).json()
    assert m["total_rows"] >= 1004  # 2 written + 1002 streamed


def test_bad_token_rejected(app_client):
    r = app_client.get("/tables", headers={"Authorization": "Bearer garbage.token"})
    assert r.status_code == 401


def test_cli_basics(catalog, tmp_path, monkeypatch):
    from click.testing import CliRunner

    import lakesoul_amd.cli as cli_mod

    monkeypatch.setattr(cli_mod, "_catalog", lambda: catalog)
    runner = CliRunner()
    r = runner.invoke(cli_mod.cli, ["create-table", "clit", "--schema", "id:int64,v:float64",
                                    "--primary-keys", "id", "--hash-bucket-num", "2"])
    assert r.exit_code == 0, r.output
    t = catalog.table("clit")
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10)})
    r = runner.invoke(cli_mod.cli, ["list-tables"])
    assert "clit" in r.output
    r = runner.invoke(cli_mod.cli, ["describe", "clit"])
    assert '"hash_bucket_num": 2' in r.output
    r = runner.invoke(cli_mod.cli, ["scan", "clit", "--limit", "5"])
    assert r.exit_code == 0 and "10 rows" in r.output
    r = runner.invoke(cli_mod.cli, ["history", "clit"])
    assert "MergeCommit" in r.output
