"""Object-store layer tests: mock:// remote scheme + disk read-through
cache + remote table roundtrip (write -> upload, scan -> localize)."""

import os

import numpy as np
import pytest

from lakesoul_amd.io.fs import DiskCache, FileSystem, is_remote
from lakesoul_amd.io.schema import Field, Schema


@pytest.fixture()
def mock_fs(tmp_path, monkeypatch):
    monkeypatch.setenv("LAKESOUL_MOCK_FS_ROOT", str(tmp_path / "remote"))
    monkeypatch.setenv("LAKESOUL_CACHE_DIR", str(tmp_path / "cache"))
    os.makedirs(tmp_path / "remote", exist_ok=True)
    import lakesoul_amd.io.fs as fsmod

    fsmod._default_fs = None  # reset singleton for env
    yield FileSystem(DiskCache(str(tmp_path / "cache")))
    fsmod._default_fs = None


def test_scheme_detection():
    assert not is_remote("/tmp/x.parquet")
    assert is_remote("mock://bucket/x.parquet")
    assert is_remote("s3://bucket/x.parquet")


def test_mock_upload_localize_roundtrip(mock_fs, tmp_path):
    src = tmp_path / "data.bin"
    src.write_bytes(b"hello lakehouse")
    mock_fs.upload(str(src), "mock://t1/data.bin")
    lp = mock_fs.localize("mock://t1/data.bin")
    assert open(lp, "rb").read() == b"hello lakehouse"
    # second localize hits the cache (same path, no re-copy)
    assert mock_fs.localize("mock://t1/data.bin") == lp


def test_cache_eviction(tmp_path):
    cache = DiskCache(str(tmp_path / "c"), capacity=1000)
    for i in range(10):
        src = tmp_path / f"f{i}"
        src.write_bytes(bytes(300))
        cache.put_from(f"mock://x/f{i}", str(src))
    assert cache.stats()["bytes"] <= 1000


def test_remote_table_roundtrip(mock_fs, tmp_path, meta_store, monkeypatch):
    """Table whose table_path is a mock:// URI: writes upload, scans pull
    through the disk cache."""
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.tables.catalog import LakeSoulCatalog

    catalog = LakeSoulCatalog(MetaClient(meta_store), warehouse=str(tmp_path / "wh"))
    t = catalog.create_table(
        "remote_t",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
        table_path="mock://warehouse/remote_t",
    )
    n = 1000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})
    t.upsert({"id": np.arange(0, n, 5, dtype=np.int64), "v": np.ones(len(range(0, n, 5)))})
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == n
    assert df["v"][::5].eq(1.0).all()
    # files live in the mock remote, not locally
    for f in t.files():
        assert f.path.startswith("mock://")


def test_retry_io_backoff_and_raise():
    """Remote IO retries with exponential backoff (reference
    object_store.rs retry config)."""
    from lakesoul_amd.io.fs import retry_io

    calls = []

    def flaky():
        calls.append(1)
        if len(calls) < 3:
            raise OSError("transient")
        return "ok"

    assert retry_io(flaky, attempts=4, base_delay=0.001) == "ok"
    assert len(calls) == 3
    calls.clear()

    def always():
        calls.append(1)
        raise OSError("down")

    import pytest as _pytest

    with _pytest.raises(OSError, match="down"):
        retry_io(always, attempts=3, base_delay=0.001)
    assert len(calls) == 3
    # non-retryable errors propagate immediately
    calls.clear()

    def typeerr():
        calls.append(1)
        raise TypeError("bug")

    with _pytest.raises(TypeError):
        retry_io(typeerr, attempts=5, base_delay=0.001)
    assert len(calls) == 1
