"""Ecosystem adapter tests: huggingface loader, torch IterableDataset
single-process, arrow dataset, ray/daft gating."""

import numpy as np
import pytest

from lakesoul_amd.io.schema import Field, Schema


@pytest.fixture()
def small_table(catalog):
    t = catalog.create_table(
        "eco",
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("s", "string")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 500
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.arange(n, dtype=np.float64),
              "s": [f"x{i}" for i in range(n)]})
    return t


def test_huggingface_loader(small_table):
    datasets = pytest.importorskip("datasets")
    from lakesoul_amd.huggingface import from_lakesoul

    ds = from_lakesoul(small_table, columns=["id", "v"])
    assert len(ds) == 500
    assert set(ds.column_names) == {"id", "v"}
    assert ds[0]["id"] in range(500)


def test_torch_iterable_dataset_single(small_table):
    from torch.utils.data import DataLoader

    from lakesoul_amd.torch.dataset import LakeSoulIterableDataset

    ds = LakeSoulIterableDataset(small_table, columns=["id", "v"], device="cpu")
    ids = []
    for item in ds:
        assert "id" in item and "v" in item
        ids.append(item["id"].numpy())
    assert sum(len(x) for x in ids) == 500


def test_arrow_dataset(small_table):
    from lakesoul_amd.arrow import LakeSoulArrowDataset

    ds = LakeSoulArrowDataset(small_table, columns=["id", "s"])
    assert ds.count_rows() == 500
    t = ds.to_table()
    assert t.num_rows == 500
    h = ds.head(7)
    assert h.num_rows == 7
    assert [f.name for f in ds.schema] == ["id", "s"]


def test_ray_daft_gated(small_table):
    import lakesoul_amd.ray as lray
    import lakesoul_amd.daft as ldaft

    with pytest.raises(ImportError, match="ray"):
        lray.read_lakesoul(small_table)
    with pytest.raises(ImportError, match="daft"):
        ldaft.read_lakesoul(small_table)


def test_torch_dataloader_workers(small_table):
    """DataLoader num_workers=2: worker_info sharding covers every row
    exactly once (reference arrow/dataset.py rank*worker sharding)."""
    import torch

    from lakesoul_amd.torch.dataset import LakeSoulIterableDataset

    ds = LakeSoulIterableDataset(small_table, columns=["id"], device="cpu")
    dl = torch.utils.data.DataLoader(
        ds, batch_size=None, num_workers=2,
        collate_fn=lambda x: x, persistent_workers=False,
    )
    ids = []
    for item in dl:
        ids.extend(item["id"].flatten().tolist())
    assert sorted(ids) == sorted(
        small_table.to_pandas()["id"].tolist())
