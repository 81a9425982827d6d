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


def test_s3_proxy_rbac(tmp_path, monkeypatch):
    fastapi = pytest.importorskip("fastapi")
    from fastapi.testclient import TestClient

    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.service.s3_proxy import create_s3_proxy

    monkeypatch.setenv("LAKESOUL_MOCK_FS_ROOT", str(tmp_path / "remote"))
    monkeypatch.setenv("LAKESOUL_CACHE_DIR", str(tmp_path / "cache"))
    import lakesoul_amd.io.fs as fsmod

    fsmod._default_fs = None
    (tmp_path / "remote").mkdir()

    catalog = LakeSoulCatalog(
        MetaClient(SqliteMetaStore(str(tmp_path / "meta.db"))),
        warehouse=str(tmp_path / "wh"),
    )
    # a table in a restricted domain at a mock:// path
    t = catalog.create_table(
        "secret_t",
        Schema([Field("id", "int64", False)]),
        table_path="mock://lake/secret_t",
    )
    t.info.domain = "teamA"
    catalog.client.store._conn().execute(
        "UPDATE table_info SET domain='teamA' WHERE table_id=?", (t.table_id,)
    )
    catalog.client.store._conn().commit()

    client = TestClient(create_s3_proxy(catalog, backend_scheme="mock", secret="s"))

    def tok(domain):
        from lakesoul_amd.service.server import TokenService

        return {"Authorization": f"Bearer {TokenService('s').issue('u', domain)}"}

    # write an object into the table path as teamA
    r = client.put("/lake/secret_t/part-abc_0000.parquet", content=b"DATA",
                   headers=tok("teamA"))
    assert r.status_code == 200
    # teamA can read it back
    r = client.get("/lake/secret_t/part-abc_0000.parquet", headers=tok("teamA"))
    assert r.status_code == 200 and r.content == b"DATA"
    # teamB is denied
    r = client.get("/lake/secret_t/part-abc_0000.parquet", headers=tok("teamB"))
    assert r.status_code == 403
    # unauthenticated is rejected
    r = client.get("/lake/secret_t/part-abc_0000.parquet")
    assert r.status_code == 401
    # objects outside any table pass through for any authenticated domain
    r = client.put("/lake/free/obj.bin", content=b"x", headers=tok("teamB"))
    assert r.status_code == 200
    r = client.get("/lake/free/obj.bin", headers=tok("teamB"))
    assert r.status_code == 200
    m = client.get("/__metrics").json()
    assert m["denied"] == 1 and m["allowed"] >= 4


def test_metrics_prometheus_format(app_client):
    hdr = _token(app_client)
    r = app_client.get("/metrics/prometheus", headers=hdr)
    assert r.status_code == 200
    body = r.text
    assert "# TYPE lakesoul_total_rows counter" in body
    assert "lakesoul_active_streams" in body


def test_sql_endpoint(app_client):
    hdr = _token(app_client)
    r = app_client.post("/sql", json={"query": "SELECT count(*) n FROM served"},
                        headers=hdr)
    assert r.status_code == 200
    body = r.json()
    assert body["columns"] == ["n"]
    assert body["rows"][0][0] >= 0
    r = app_client.post("/sql", json={"query": "BOGUS"}, headers=hdr)
    assert r.status_code == 400


def test_s3_proxy_prometheus_metrics(tmp_path, monkeypatch, catalog):
    from fastapi.testclient import TestClient

    from lakesoul_amd.service.s3_proxy import create_s3_proxy

    monkeypatch.setenv("LAKESOUL_MOCK_FS_ROOT", str(tmp_path / "store"))
    (tmp_path / "store").mkdir()
    app = create_s3_proxy(catalog)
    c = TestClient(app)
    r = c.get("/__metrics/prometheus")
    assert r.status_code == 200
    assert "# TYPE lakesoul_s3proxy_" in r.text


def test_table_splits_endpoint(app_client, catalog):
    import numpy as np

    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "gsp", Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=2)
    t.upsert({"id": np.arange(8, dtype=np.int64), "v": np.zeros(8)})
    tok = app_client.post("/handshake", json={"username": "u"}).json()["token"]
    r = app_client.get("/table/gsp/splits",
                       headers={"Authorization": f"Bearer {tok}"})
    assert r.status_code == 200, r.text
    splits = r.json()["splits"]
    assert sorted(s["hash_bucket"] for s in splits) == [0, 1]
    for s in splits:
        assert s["primary_keys"] == ["id"] and len(s["file_paths"]) == 1
