"""Auxiliary subsystem tests: streaming writer, CDC ingestion with
exactly-once checkpoints, auto-compaction service, table stats."""

import numpy as np
import pytest

from lakesoul_amd.io.schema import Field, Schema


def test_streaming_writer_commit(catalog):
    from lakesoul_amd.io.stream_writer import StreamingWriter

    t = catalog.create_table(
        "sw",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    with StreamingWriter(t, max_rows_per_flush=500) as w:
        for i in range(4):
            w.write({"id": np.arange(i * 300, (i + 1) * 300, dtype=np.int64),
                     "v": np.full(300, float(i))})
    assert t.scan().count() == 1200
    # single commit version despite multiple flushes
    assert t.latest_version() == 0


def test_streaming_writer_abort_leaves_nothing(catalog):
    from lakesoul_amd.io.stream_writer import StreamingWriter

    t = catalog.create_table(
        "swa",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
    )
    try:
        with StreamingWriter(t, max_rows_per_flush=10) as w:
            w.write({"id": np.arange(50, dtype=np.int64), "v": np.zeros(50)})
            raise RuntimeError("boom")
    except RuntimeError:
        pass
    assert t.scan().count() == 0
    assert t.files() == []


def test_cdc_ingestor_exactly_once(catalog):
    from lakesoul_amd.ingest.cdc import CdcIngestor

    t = catalog.create_table(
        "cdct",
        Schema([Field("id", "int64", False), Field("v", "float64"), Field("rk", "string")]),
        primary_keys=["id"],
        properties={"lakesoul_cdc_change_column": "rk"},
    )
    events = [
        {"op": "insert", "data": {"id": 1, "v": 1.0}, "offset": 0},
        {"op": "insert", "data": {"id": 2, "v": 2.0}, "offset": 1},
        {"op": "update", "data": {"id": 1, "v": 9.0}, "offset": 2},
        {"op": "delete", "data": {"id": 2, "v": 0.0}, "offset": 3},
    ]
    ing = CdcIngestor(t, "src1", checkpoint_rows=100)
    assert ing.ingest(events) == 4
    ing.close()
    df = t.to_pandas()
    assert df["id"].tolist() == [1] and df["v"].tolist() == [9.0]
    assert ing.committed_offset() == 3
    # replay the same events: nothing applied (exactly-once)
    ing2 = CdcIngestor(t, "src1", checkpoint_rows=100)
    assert ing2.ingest(events) == 0
    ing2.close()
    assert len(t.to_pandas()) == 1
    # new events continue
    assert ing2.ingest([{"op": "insert", "data": {"id": 3, "v": 3.0}, "offset": 4}]) == 1
    ing2.close()
    assert sorted(t.to_pandas()["id"].tolist()) == [1, 3]


def test_compaction_service_trigger(catalog):
    from lakesoul_amd.service.compactor import CompactionService

    t = catalog.create_table(
        "autoc",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=1,
    )
    svc = CompactionService(catalog)
    assert svc.scan_once() == []  # nothing yet
    for i in range(11):  # versions 0..10 -> trigger
        t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.full(10, float(i))})
    done = svc.scan_once()
    assert ("autoc", "-5") in done
    assert all("compactdir" in f.path for f in t.files())
    df = t.to_pandas()
    assert df["v"].tolist() == [10.0] * 10
    # second cycle: no trigger (just compacted)
    assert svc.scan_once() == []


def test_table_stats(catalog):
    from lakesoul_amd.tables.stats import table_stats

    t = catalog.create_table(
        "statt",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    t.upsert({"id": np.arange(100, dtype=np.int64), "v": np.zeros(100)})
    t.upsert({"id": np.arange(50, dtype=np.int64), "v": np.ones(50)})
    s = table_stats(t)
    assert s.file_count == 4  # 2 buckets x 2 commits
    assert s.total_bytes > 0
    assert s.partitions[0].version == 1


def test_hbm_scan_cache(catalog, monkeypatch):
    from lakesoul_amd.io.hbm_cache import scan_cache

    scan_cache().clear()
    t = catalog.create_table(
        "cached",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=2,
    )
    n = 1000
    t.upsert({"id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})
    t.upsert({"id": np.arange(100, dtype=np.int64), "v": np.ones(100)})

    import pandas as pd

    df1 = t.scan(device="cpu", options={"scan_cache": "1"}).to_arrow().to_pandas()
    s0 = scan_cache().stats()
    assert s0["entries"] == 2 and s0["misses"] >= 2
    df2 = t.scan(device="cpu", options={"scan_cache": "1"}).to_arrow().to_pandas()
    s1 = scan_cache().stats()
    assert s1["hits"] >= 2
    pd.testing.assert_frame_equal(
        df1.sort_values("id").reset_index(drop=True),
        df2.sort_values("id").reset_index(drop=True),
    )
    # a new commit changes the key -> fresh read, correct data
    t.upsert({"id": np.array([5], dtype=np.int64), "v": np.array([42.0])})
    df3 = t.scan(device="cpu", options={"scan_cache": "1"}).to_arrow().to_pandas()
    assert df3.loc[df3["id"] == 5, "v"].iloc[0] == 42.0


def test_cleanup_old_versions(catalog):
    t = catalog.create_table(
        "vac",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"],
        hash_bucket_num=1,
    )
    for i in range(5):
        t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.full(10, float(i))})
    t.compaction()  # compacted snapshot replaces the deltas
    import os

    all_files_before = [f.path for f in t.files()]
    removed = t.cleanup_old_versions(keep_latest=1)
    assert removed >= 5  # old delta files physically gone
    df = t.to_pandas()
    assert df["v"].tolist() == [4.0] * 10
    # old version no longer resolvable
    assert t.client.files_for_partition(t.table_id, "-5", version=0) == []
    for f in t.files():
        assert os.path.exists(f.path)


def test_vacuum_is_atomic_under_concurrent_commit(catalog):
    """cleanup_old_versions must never expose an empty version history:
    a commit racing the vacuum lands on top of the kept versions."""
    import threading

    t = catalog.create_table(
        "vac2",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=1,
    )
    for it in range(6):
        t.upsert({"id": np.arange(10, dtype=np.int64),
                  "v": np.full(10, float(it))})
    stop = threading.Event()
    errs = []

    def committer():
        i = 100
        while not stop.is_set():
            try:
                t.upsert({"id": np.array([i], dtype=np.int64),
                          "v": np.array([1.0])})
                i += 1
            except Exception as e:  # pragma: no cover
                errs.append(repr(e))
                return

    th = threading.Thread(target=committer)
    th.start()
    try:
        for _ in range(4):
            t.cleanup_old_versions(keep_latest=2)
    finally:
        stop.set()
        th.join()
    assert not errs, errs
    df = t.to_pandas()
    assert len(df) >= 10
    assert (df[df.id < 10]["v"] == 5.0).all()  # newest base survives


def test_timing_phases(monkeypatch):
    from lakesoul_amd.utils import timing

    monkeypatch.setattr(timing, "ENABLED", True)
    timing.reset()
    with timing.phase("unit_test_phase"):
        pass
    rep = timing.report()
    assert "unit_test_phase" in rep
    timing.reset()
    assert "unit_test_phase" not in timing.report()


def test_ioconfig_env_fallback(monkeypatch):
    """Option map falls back to LAKESOUL_<KEY> env (reference
    config/mod.rs:160-165 env fallback)."""
    from lakesoul_amd.config import IOConfig

    monkeypatch.setenv("LAKESOUL_MY_CUSTOM_OPT", "hello")
    cfg = IOConfig()
    assert cfg.option("my_custom_opt", "fallback") == "hello"
    assert cfg.option("absent_opt", "fallback") == "fallback"


def test_notify_driven_compaction(catalog):
    """Notify-driven compaction pipeline (reference meta_init.sql:102-150
    trigger -> pg_notify -> listener): the local bus fires when a
    partition crosses 10 delta commits and the subscribed compactor
    compacts exactly that partition."""
    import numpy as np

    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.meta.entities import CommitOp
    from lakesoul_amd.meta.notify import LocalNotifyBus, NotifyDrivenCompactor

    bus = LocalNotifyBus()
    catalog.client.notify_bus = bus
    try:
        t = catalog.create_table(
            "notif",
            Schema([Field("id", "int64", False), Field("v", "float64")]),
            primary_keys=["id"], hash_bucket_num=1,
        )
        comp = NotifyDrivenCompactor(catalog, bus)
        for i in range(10):
            t.upsert({"id": np.arange(10, dtype=np.int64),
                      "v": np.full(10, float(i))})
        assert not bus.published  # below the threshold: no event yet
        assert comp.drain() == 0
        t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.full(10, 99.0)})
        assert len(bus.published) == 1
        ev = bus.published[0]
        assert ev.table_id == t.table_id and ev.partition_desc == "-5"
        # payload round trip (what pg_notify would carry)
        from lakesoul_amd.meta.notify import CompactionEvent

        assert CompactionEvent.from_payload(ev.payload()) == ev
        assert comp.drain() == 1
        # compaction committed: latest version is a CompactionCommit and
        # data still reads correctly
        cur = catalog.client.store.get_latest_partition_info(t.table_id, "-5")
        assert cur.commit_op is CommitOp.CompactionCommit
        df = t.to_pandas()
        assert len(df) == 10 and (df["v"] == 99.0).all()
        # further single delta does not re-fire
        t.upsert({"id": np.arange(10, dtype=np.int64), "v": np.zeros(10)})
        assert len(bus.published) == 1
    finally:
        catalog.client.notify_bus = None


def test_memprof_snapshot_and_track(capsys):
    """Memory accounting hooks (reference mem.rs jemalloc prof +
    LoggedMemoryPool analog)."""
    from lakesoul_amd.utils import memprof

    s = memprof.snapshot()
    assert s["rss"] > 0 and s["peak_rss"] >= s["rss"]
    assert "hbm_allocated" in s
    msgs = []
    with memprof.track("alloc-test", log=msgs.append):
        buf = bytearray(50 * 1024 * 1024)
        assert len(buf) == 50 * 1024 * 1024
    assert len(msgs) == 1 and "alloc-test" in msgs[0]
