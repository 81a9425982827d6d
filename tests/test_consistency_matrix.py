"""Table-level writer x reader consistency matrix.

The reference's contract test for "same table readable everywhere" is
``python/tests/compat/run_matrix.py``: every engine
writes each case table, every other engine reads it, results are
normalized and diffed. Here the engines are this framework's own
surfaces: writers = {client upsert, SQL INSERT, StreamingWriter},
readers = {python scan, SQL SELECT, C-ABI reader over the snapshot
files, HTTP gateway Arrow-IPC stream}. Every combination must produce
the identical normalized table."""

import ctypes
import io as _io
import os

import numpy as np
import pandas as pd
import pytest

from lakesoul_amd.io.schema import Field, Schema

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
LIB = os.path.join(REPO, "lakesoul_amd", "liblakesoul_amd_c.so")

ROWS1 = {
    "id": np.arange(20, dtype=np.int64),
    "v": np.where(np.arange(20) % 5 == 0, np.nan, np.arange(20) * 1.5),
    "s": [None if i % 7 == 0 else f"s{i}" for i in range(20)],
}
UPD = {
    "id": np.array([1, 3, 5], dtype=np.int64),
    "v": np.array([100.0, 300.0, 500.0]),
    "s": ["u1", "u3", "u5"],
}


def _schema():
    return Schema([Field("id", "int64", False), Field("v", "float64"),
                   Field("s", "string")])


def _normalize(df: pd.DataFrame) -> pd.DataFrame:
    df = df[["id", "v", "s"]].sort_values("id").reset_index(drop=True)
    df["id"] = df["id"].astype("int64")
    df["v"] = df["v"].astype("float64")
    df["s"] = df["s"].astype("object").where(df["s"].notna(), None)
    return df


def _expected() -> pd.DataFrame:
    df = pd.DataFrame({"id": ROWS1["id"], "v": ROWS1["v"],
                       "s": ROWS1["s"]})
    for i, rid in enumerate(UPD["id"]):
        df.loc[df["id"] == rid, "v"] = UPD["v"][i]
        df.loc[df["id"] == rid, "s"] = UPD["s"][i]
    return _normalize(df)


# ---------------------------------------------------------------- writers

def write_client(catalog, name):
    t = catalog.create_table(name, _schema(), primary_keys=["id"],
                             hash_bucket_num=2)
    t.upsert(ROWS1)
    t.upsert(UPD)
    return t


def write_sql(catalog, name):
    from lakesoul_amd.sql import execute_sql

    execute_sql(catalog, f"CREATE TABLE {name} (id BIGINT NOT NULL, "
                         "v DOUBLE, s VARCHAR(16)) PRIMARY KEY (id) "
                         "HASH BUCKETS 2")

    def vals(rows):
        out = []
        for i in range(len(rows["id"])):
            v = rows["v"][i]
            s = rows["s"][i] if isinstance(rows["s"], list) else rows["s"][i]
            vtxt = "NULL" if (isinstance(v, float) and np.isnan(v)) else repr(float(v))
            stxt = "NULL" if s is None else f"'{s}'"
            out.append(f"({int(rows['id'][i])}, {vtxt}, {stxt})")
        return ", ".join(out)

    execute_sql(catalog, f"INSERT INTO {name} VALUES {vals(ROWS1)}")
    execute_sql(catalog, f"INSERT INTO {name} VALUES {vals(UPD)}")
    return catalog.table(name)


def write_streaming(catalog, name):
    from lakesoul_amd.io.stream_writer import StreamingWriter

    t = catalog.create_table(name, _schema(), primary_keys=["id"],
                             hash_bucket_num=2)
    with StreamingWriter(t, max_rows_per_flush=7) as w:
        w.write(ROWS1)
    with StreamingWriter(t, max_rows_per_flush=7) as w:
        w.write(UPD)
    return t


WRITERS = [("client", write_client), ("sql", write_sql),
           ("streaming", write_streaming)]


# ---------------------------------------------------------------- readers

def read_scan(catalog, t):
    return t.scan().to_arrow().to_pandas()


def read_sql(catalog, t):
    from lakesoul_amd.sql import execute_sql

    return execute_sql(catalog, f"SELECT * FROM {t.info.table_name}")


def read_capi(catalog, t):
    """C-ABI reader: snapshot files from the metadata layer, MOR merge
    configured with the table's PKs (what a JVM connector does)."""
    if not os.path.exists(LIB):
        pytest.skip("liblakesoul_amd_c.so not built")
    import pyarrow as pa

    L = ctypes.CDLL(LIB)
    vp, cp, i64 = ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int64
    L.lakesoul_c_config_create.restype = vp
    L.lakesoul_c_config_add_file.argtypes = [vp, cp]
    L.lakesoul_c_config_add_primary_key.argtypes = [vp, cp]
    L.lakesoul_c_config_free.argtypes = [vp]
    L.lakesoul_c_reader_create_from_config.restype = vp
    L.lakesoul_c_reader_create_from_config.argtypes = [vp]
    L.lakesoul_c_reader_start.argtypes = [vp]
    L.lakesoul_c_reader_schema.argtypes = [vp, vp]
    L.lakesoul_c_reader_next.argtypes = [vp, vp]
    L.lakesoul_c_reader_close.argtypes = [vp]
    L.lakesoul_c_last_error.restype = cp

    frames = []
    for unit in t.scan().plan():
        if not unit.files:
            continue
        cfg = ctypes.c_void_p(L.lakesoul_c_config_create())
        for f in unit.files:
            L.lakesoul_c_config_add_file(cfg, f.encode())
        for p in t.primary_keys:
            L.lakesoul_c_config_add_primary_key(cfg, p.encode())
        r = L.lakesoul_c_reader_create_from_config(cfg)
        L.lakesoul_c_config_free(cfg)
        assert r, L.lakesoul_c_last_error()
        assert L.lakesoul_c_reader_start(ctypes.c_void_p(r)) == 0, \
            L.lakesoul_c_last_error()
        sh = (ctypes.c_byte * 512)()
        assert L.lakesoul_c_reader_schema(
            ctypes.c_void_p(r), ctypes.addressof(sh)) == 0
        schema = pa.Schema._import_from_c(ctypes.addressof(sh))
        while True:
            ah = (ctypes.c_byte * 512)()
            rc = L.lakesoul_c_reader_next(ctypes.c_void_p(r),
                                          ctypes.addressof(ah))
            assert rc >= 0, L.lakesoul_c_last_error()
            if rc == 0:
                break
            arr = pa.Array._import_from_c(ctypes.addressof(ah),
                                          pa.struct(list(schema)))
            frames.append(pa.Table.from_struct_array(arr).to_pandas())
        L.lakesoul_c_reader_close(ctypes.c_void_p(r))
    return pd.concat(frames, ignore_index=True)


def read_gateway(catalog, t):
    """HTTP gateway: /table/{name}/scan streaming Arrow IPC."""
    fastapi = pytest.importorskip("fastapi")
    import pyarrow.ipc as ipc
    from fastapi.testclient import TestClient

    from lakesoul_amd.service.server import create_app

    client = TestClient(create_app(catalog, secret="cm"))
    tok = client.post("/handshake", json={"username": "m"}).json()["token"]
    resp = client.get(f"/table/{t.info.table_name}/scan",
                      headers={"Authorization": f"Bearer {tok}"})
    assert resp.status_code == 200, resp.text
    reader = ipc.open_stream(_io.BytesIO(resp.content))
    return reader.read_all().to_pandas()


READERS = [("scan", read_scan), ("sql", read_sql), ("capi", read_capi),
           ("gateway", read_gateway)]


@pytest.mark.parametrize("wname,writer", WRITERS, ids=[w[0] for w in WRITERS])
def test_matrix_pk_mor(catalog, wname, writer):
    """Every reader sees the identical normalized MOR result no matter
    which writer produced the table."""
    t = writer(catalog, f"cm_{wname}")
    exp = _expected()
    for rname, reader in READERS:
        got = _normalize(reader(catalog, t))
        pd.testing.assert_frame_equal(got, exp, check_dtype=True), \
            (wname, rname)


def test_matrix_nested_columns(catalog):
    """Nested columns (list / list<string> / struct / map) round the
    writer->scan/sql/gateway matrix."""
    schema = Schema([
        Field("id", "int64", False),
        Field("emb", "list<float32>"),
        Field("tags", "list<string>"),
        Field("st", "struct<a:int64,b:string>"),
        Field("mp", "map<string,int64>"),
    ])
    t = catalog.create_table("cm_nested", schema, primary_keys=["id"],
                             hash_bucket_num=2)
    rows = {
        "id": np.arange(8, dtype=np.int64),
        "emb": [None if i % 3 == 0 else [float(i), i + 0.5]
                for i in range(8)],
        "tags": [None if i % 4 == 0 else [f"t{i}", ""] for i in range(8)],
        "st": [None if i % 5 == 0 else {"a": i, "b": f"b{i}"}
               for i in range(8)],
        "mp": [None if i % 2 == 0 else {f"k{i}": i} for i in range(8)],
    }
    t.upsert(rows)
    t.upsert({"id": np.array([1], dtype=np.int64), "emb": [[9.0]],
              "tags": [["z"]], "st": [{"a": -1, "b": "u"}],
              "mp": [{"u": 0}]})
    base = t.scan().to_arrow().sort_by("id")
    gw = read_gateway(catalog, t)
    gw = gw.sort_values("id").reset_index(drop=True)
    bp = base.to_pandas()
    for cname in ("emb", "tags", "st", "mp"):
        a = [None if x is None else x for x in bp[cname]]
        b = [None if x is None else x for x in gw[cname]]
        for x, y in zip(a, b):
            if x is None or (isinstance(x, float) and pd.isna(x)):
                assert y is None or (isinstance(y, float) and pd.isna(y))
            else:
                assert list(x) == list(y), (cname, x, y)
    assert base.column("st").to_pylist()[1] == {"a": -1, "b": "u"}


def test_matrix_cdc_delete(catalog):
    """CDC delete semantics agree between scan and SQL readers."""
    t = catalog.create_table(
        "cm_cdc",
        Schema([Field("id", "int64", False), Field("v", "float64"),
                Field("rowKinds", "string")]),
        primary_keys=["id"],
        properties={"lakesoul_cdc_change_column": "rowKinds"})
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.ones(10),
              "rowKinds": ["insert"] * 10})
    t.upsert({"id": np.array([2, 8], dtype=np.int64), "v": np.zeros(2),
              "rowKinds": ["delete", "delete"]})
    from lakesoul_amd.sql import execute_sql

    a = t.scan().to_arrow().to_pandas().sort_values("id")["id"].tolist()
    b = execute_sql(catalog, "SELECT id FROM cm_cdc").sort_values("id")[
        "id"].tolist()
    assert a == b == [0, 1, 3, 4, 5, 6, 7, 9]
