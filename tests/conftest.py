import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: requires an MI355X GPU (run on gpurun box)")


@pytest.fixture()
def meta_store(tmp_path):
    from lakesoul_amd.meta.store import SqliteMetaStore

    return SqliteMetaStore(str(tmp_path / "meta.db"))


@pytest.fixture()
def catalog(tmp_path, meta_store):
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.tables.catalog import LakeSoulCatalog

    return LakeSoulCatalog(MetaClient(meta_store), warehouse=str(tmp_path / "wh"))
