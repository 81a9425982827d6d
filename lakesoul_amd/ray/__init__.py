"""Ray datasource/datasink (reference: ``python/src/lakesoul/ray/``).

Ray is not installed in this build image (no network); the adapters are
importable and raise a clear error on use without ray. With ray present:

    import ray
    from lakesoul_amd.ray import read_lakesoul, write_lakesoul
    ds = read_lakesoul(table)            # ray.data.Dataset over scan units
    write_lakesoul(ds, table)            # distributed upsert + one commit
"""

from __future__ import annotations

from typing import Optional, Sequence


def _require_ray():
    try:
        import ray  # noqa: F401
        import ray.data  # noqa: F401

        return ray
    except ImportError as e:
        raise ImportError(
            "ray is not installed in this environment; "
            "lakesoul_amd.ray needs the 'ray[data]' package"
        ) from e


def read_lakesoul(table, columns: Optional[Sequence[str]] = None, filters=None,
                  parallelism: int = -1):
    """ray.data.Dataset over the table's scan units (each unit = one
    merge-on-read task, matching the reference's per-split reads)."""
    ray = _require_ray()
    scan = table.scan(columns=columns, filters=filters, device="cpu")
    units = scan.plan()
    table_name, namespace = table.info.table_name, table.info.table_namespace
    meta_db = table.client.store.path

    def read_unit(unit_idx: int):
        import os

        os.environ["LAKESOUL_META_DB"] = meta_db
        from lakesoul_amd.meta.client import MetaClient
        from lakesoul_amd.meta.store import SqliteMetaStore
        from lakesoul_amd.tables.catalog import LakeSoulCatalog

        cat = LakeSoulCatalog(MetaClient(SqliteMetaStore(meta_db)))
        t = cat.table(table_name, namespace)
        s = t.scan(columns=columns, filters=filters, device="cpu")
        us = s.plan()
        return [s._read_unit(us[unit_idx]).to_arrow()]

    return ray.data.from_items(list(range(len(units)))).flat_map(
        lambda i: [{"__unit": i}]
    ).map_batches(lambda b: read_unit(int(b["__unit"][0])))


def write_lakesoul(ds, table) -> None:
    """Write a ray.data.Dataset into the table (per-block writes, single
    metadata commit via StreamingWriter)."""
    _require_ray()
    from lakesoul_amd.io.stream_writer import StreamingWriter

    with StreamingWriter(table) as w:
        for batch in ds.iter_batches(batch_format="pyarrow"):
            w.write(batch)
