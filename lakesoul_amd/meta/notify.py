"""Compaction notification bus — the reference's PG trigger/notify
pipeline (``script/meta_init.sql:102-150``: the partition_insert trigger
fires pg_notify('lakesoul_compaction_notify', payload) once a partition
accumulates >= 10 delta commits; Flink listeners wake on the channel).

Two transports behind one seam:

- ``LocalNotifyBus``: in-process pub/sub fired by the metadata client on
  every commit that crosses the delta threshold — the single-node
  deployment (SqliteMetaStore) equivalent of the trigger.
- ``PgNotifyBus``: LISTEN on the reference channel over a psycopg
  connection (import-gated like PostgresMetaStore; the trigger itself
  ships with the reference schema and needs no work on our side).

The CompactionService subscribes instead of polling; polls remain as a
fallback sweep for missed events.
"""

from __future__ import annotations

import json
import queue
import threading
from dataclasses import dataclass
from typing import Callable, List, Optional

COMPACTION_CHANNEL = "lakesoul_compaction_notify"
COMPACTION_DELTA_THRESHOLD = 10  # meta_init.sql trigger rule


@dataclass
class CompactionEvent:
    table_id: str
    partition_desc: str
    version: int

    def payload(self) -> str:
        # the reference trigger sends a json payload with table/partition
        return json.dumps({
            "table_id": self.table_id,
            "partition_desc": self.partition_desc,
            "version": self.version,
        })

    @classmethod
    def from_payload(cls, s: str) -> "CompactionEvent":
        d = json.loads(s)
        return cls(d["table_id"], d["partition_desc"], int(d.get("version", 0)))


class LocalNotifyBus:
    """In-process pub/sub with the PG trigger's threshold rule applied by
    the publisher (meta client)."""

    def __init__(self):
        self._subs: List[Callable[[CompactionEvent], None]] = []
        self._lock = threading.Lock()
        self.published: List[CompactionEvent] = []

    def subscribe(self, cb: Callable[[CompactionEvent], None]) -> None:
        with self._lock:
            self._subs.append(cb)

    def publish(self, ev: CompactionEvent) -> None:
        with self._lock:
            subs = list(self._subs)
            self.published.append(ev)
        for cb in subs:
            cb(ev)


class PgNotifyBus:  # pragma: no cover — needs a live PG + psycopg
    """LISTEN lakesoul_compaction_notify on a dedicated connection."""

    def __init__(self, url: str):
        try:
            import psycopg
        except ImportError as e:
            raise ImportError("PgNotifyBus needs psycopg") from e
        self._conn = psycopg.connect(url, autocommit=True)
        self._conn.execute(f"LISTEN {COMPACTION_CHANNEL}")
        self._subs: List[Callable[[CompactionEvent], None]] = []
        self._stop = threading.Event()
        self._t = threading.Thread(target=self._loop, daemon=True)
        self._t.start()

    def subscribe(self, cb) -> None:
        self._subs.append(cb)

    def _loop(self):
        gen = self._conn.notifies()
        for note in gen:
            if self._stop.is_set():
                return
            try:
                ev = CompactionEvent.from_payload(note.payload)
            except Exception:
                continue
            for cb in list(self._subs):
                cb(ev)

    def close(self):
        self._stop.set()
        self._conn.close()


class NotifyDrivenCompactor:
    """Wakes on compaction events instead of polling; a periodic sweep
    still covers events missed while down (the reference pairs the
    notify listener with catch-up scans the same way)."""

    def __init__(self, catalog, bus, device: Optional[str] = None):
        self.catalog = catalog
        self.bus = bus
        self.device = device
        self._q: "queue.Queue[CompactionEvent]" = queue.Queue()
        self.compacted: List[CompactionEvent] = []
        bus.subscribe(self._q.put)

    def drain(self, timeout: float = 0.0) -> int:
        """Process queued events; returns how many partitions compacted."""
        n = 0
        while True:
            try:
                ev = self._q.get(timeout=timeout) if timeout else self._q.get_nowait()
            except queue.Empty:
                return n
            t = self._table_by_id(ev.table_id)
            if t is None:
                continue
            if t.client.compaction_needed(t.table_id, ev.partition_desc):
                t.compaction(ev.partition_desc, device=self.device)
                self.compacted.append(ev)
                n += 1

    def _table_by_id(self, table_id: str):
        for ns in self.catalog.list_namespaces():
            for name in self.catalog.list_tables(ns):
                t = self.catalog.table(name, ns)
                if t.table_id == table_id:
                    return t
        return None
