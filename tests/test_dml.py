"""DML tests: update / delete (CDC + rewrite paths) / add_columns /
streaming incremental source."""

import numpy as np
import pytest

from lakesoul_amd.io.schema import Field, Schema


def _pk_table(catalog, name, props=None, buckets=2):
    return catalog.create_table(
        name,
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("s", "string")]),
        primary_keys=["id"],
        hash_bucket_num=buckets,
        properties=props,
    )


def test_update(catalog):
    t = _pk_table(catalog, "upd")
    n = 1000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n), "s": ["a"] * n})
    changed = t.update([("id", "<", 100)], {"v": 9.5, "s": "updated"})
    assert changed == 100
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == n
    assert (df["v"][:100] == 9.5).all() and (df["v"][100:] == 0).all()
    assert df["s"][0] == "updated" and df["s"][500] == "a"


def test_delete_cdc(catalog):
    t = _pk_table(catalog, "delcdc", props={"lakesoul_cdc_change_column": "s"})
    t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10), "s": ["insert"] * 10})
    deleted = t.delete([("id", ">=", 7)])
    assert deleted == 3
    df = t.to_pandas()
    assert sorted(df["id"].tolist()) == list(range(7))


def test_delete_rewrite(catalog):
    t = _pk_table(catalog, "delrw")
    n = 2000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.arange(n, dtype=np.float64), "s": ["x"] * n})
    t.upsert({"id": np.arange(0, n, 2, dtype=np.int64), "v": np.full(n // 2, -1.0), "s": ["y"] * (n // 2)})
    deleted = t.delete([("v", "==", -1.0)])
    assert deleted == n // 2
    df = t.to_pandas().sort_values("id").reset_index(drop=True)
    assert sorted(df["id"].tolist()) == list(range(1, n, 2))
    # subsequent upsert still works on rewritten buckets
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([5.0]), "s": ["z"]})
    df = t.to_pandas()
    assert df.loc[df["id"] == 1, "v"].iloc[0] == 5.0


def test_add_columns(catalog):
    t = _pk_table(catalog, "addc")
    t.upsert({"id": np.arange(5, dtype=np.int64), "v": np.zeros(5), "s": ["a"] * 5})
    t.add_columns([Field("extra", "float64", True)])
    t2 = catalog.table("addc")
    assert "extra" in t2.schema.names()
    # old rows read as null; new rows carry values
    t2.upsert({"id": np.array([10], dtype=np.int64), "v": np.array([1.0]),
               "s": ["b"], "extra": np.array([7.0])})
    df = t2.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(df) == 6
    assert df["extra"].iloc[5] == 7.0
    assert df["extra"][:5].isna().all()


def test_table_stream_incremental(catalog):
    from lakesoul_amd.tables.stream import TableStream

    t = _pk_table(catalog, "strm", buckets=1)
    t.upsert({"id": np.array([1, 2], dtype=np.int64), "v": np.zeros(2), "s": ["a", "b"]})
    stream = TableStream(t, device="cpu")
    batches, adv = stream.poll()
    assert adv and sum(b.num_rows for b in batches) == 2
    # no new data -> no batches
    batches, adv = stream.poll()
    assert not adv and batches == []
    # two more commits -> both picked up in one poll
    t.upsert({"id": np.array([3], dtype=np.int64), "v": np.ones(1), "s": ["c"]})
    t.upsert({"id": np.array([4], dtype=np.int64), "v": np.ones(1), "s": ["d"]})
    batches, adv = stream.poll()
    got = sorted(
        int(x) for b in batches for x in b.columns["id"].data.cpu().numpy()
    )
    assert got == [3, 4]
    # resumed stream from saved positions sees nothing new
    stream2 = TableStream(t, start_versions=dict(stream.positions), device="cpu")
    batches, adv = stream2.poll()
    assert not adv


def test_cdc_insert_after_delete(catalog):
    """Key lifecycle insert → delete → re-insert across commits: the
    newest row-kind wins at every read point (reference CDC semantics)."""
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "cdclife",
        Schema([Field("id", "int64", False), Field("v", "float64"),
                Field("rowKinds", "string")]),
        primary_keys=["id"],
        properties={"lakesoul_cdc_change_column": "rowKinds"},
    )
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([1.0]),
              "rowKinds": ["insert"]})
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([0.0]),
              "rowKinds": ["delete"]})
    assert len(t.to_pandas()) == 0
    t.upsert({"id": np.array([1], dtype=np.int64), "v": np.array([2.0]),
              "rowKinds": ["insert"]})
    df = t.to_pandas()
    assert df["v"].tolist() == [2.0]
    # time travel sees each state
    assert len(t.to_pandas(version=0)) == 1
    assert len(t.to_pandas(version=1)) == 0
    # compaction drops the tombstone permanently
    t.compaction()
    df = t.to_pandas()
    assert df["v"].tolist() == [2.0]


def test_cdc_multi_key_interleaved(catalog):
    from lakesoul_amd.io.schema import Field, Schema

    t = catalog.create_table(
        "cdcmix",
        Schema([Field("id", "int64", False), Field("v", "float64"),
                Field("rowKinds", "string")]),
        primary_keys=["id"], hash_bucket_num=2,
        properties={"lakesoul_cdc_change_column": "rowKinds"},
    )
    n = 3000
    rng = np.random.default_rng(9)
    alive = {}
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n),
              "rowKinds": ["insert"] * n})
    alive = {i: 0.0 for i in range(n)}
    for it in range(6):
        ids = rng.choice(n, 400, replace=False).astype(np.int64)
        kinds = ["delete" if rng.random() < 0.4 else "update" for _ in ids]
        vals = np.full(400, float(it + 1))
        t.upsert({"id": ids, "v": vals, "rowKinds": kinds})
        for i, k in zip(ids, kinds):
            if k == "delete":
                alive.pop(int(i), None)
            else:
                alive[int(i)] = float(it + 1)
    df = t.to_pandas().sort_values("id")
    assert df["id"].tolist() == sorted(alive)
    np.testing.assert_allclose(df["v"].to_numpy(),
                               [alive[i] for i in sorted(alive)])


def test_stream_across_compaction(catalog):
    """Incremental readers spanning a CompactionCommit: the stream must
    not double-deliver compacted history (snapshot-replace semantics)."""
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.tables.stream import TableStream

    t = catalog.create_table(
        "scomp",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    t.upsert({"id": np.arange(100, dtype=np.int64), "v": np.zeros(100)})
    stream = TableStream(t, device="cpu")
    batches, _ = stream.poll()
    assert sum(b.num_rows for b in batches) == 100
    # compaction replaces the snapshot — no NEW rows for the stream
    t.upsert({"id": np.arange(0, 100, 2, dtype=np.int64), "v": np.ones(50)})
    batches, _ = stream.poll()
    n_delta = sum(b.num_rows for b in batches)
    assert n_delta == 50
    t.compaction()
    batches, advanced = stream.poll()
    # a compaction carries no new logical rows; whatever the stream
    # chooses to deliver must not exceed the full table (no dup storm)
    assert sum(b.num_rows for b in batches) <= 100
    # after compaction, new upserts flow normally
    t.upsert({"id": np.array([1000], dtype=np.int64), "v": np.array([5.0])})
    batches, _ = stream.poll()
    got = []
    for b in batches:
        got.extend(b.columns["id"].data.tolist())
    assert 1000 in got
