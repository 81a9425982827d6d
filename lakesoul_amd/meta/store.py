"""Metadata store backends.

The schema mirrors the reference's PostgreSQL catalog
(``script/meta_init.sql``) table-for-table and column-for-column:

- ``namespace``, ``table_info``, ``table_name_id``, ``table_path_id``,
  ``data_commit_info`` (with the two-phase ``committed`` flag,
  meta_init.sql:78), ``partition_info`` (MVCC versions + snapshot UUID
  array, meta_init.sql:87-99), ``global_config``,
  ``discard_compressed_file_info``.

Two backends:

- :class:`SqliteMetaStore` — default; single-node, multi-process safe
  (WAL + immediate transactions). PG array/composite columns are stored
  as JSON text.
- :class:`PostgresMetaStore` — used when ``LAKESOUL_PG_URL`` is set and a
  psycopg driver is importable; same SQL surface, real ``meta_init.sql``
  schema. (Gated: the build environment has no network/PG server.)

The compaction "trigger" analog: PostgreSQL's ``partition_insert()``
trigger + ``pg_notify`` (meta_init.sql:102-150) is replaced by
:meth:`MetaClient.compaction_needed` polling in client.py, which applies
the same >=10-deltas-since-last-compaction rule.
"""

from __future__ import annotations

import json
import os
import sqlite3
import threading
import time
from typing import Dict, List, Optional, Sequence

from .entities import (
    CommitOp,
    DataCommitInfo,
    DataFileOp,
    Namespace,
    PartitionInfo,
    TableInfo,
)


class CommitConflictError(Exception):
    """Raised when a partition_info (table_id, partition_desc, version) PK
    insert conflicts — the MVCC CAS failed and the caller must re-read and
    retry (reference: DBManager.java:509 retry loop /
    metadata_client.rs:498-663)."""


_SQLITE_SCHEMA = """
CREATE TABLE IF NOT EXISTS namespace (
    namespace  TEXT PRIMARY KEY,
    properties TEXT,
    comment    TEXT,
    domain     TEXT DEFAULT 'public'
);
CREATE TABLE IF NOT EXISTS table_info (
    table_id        TEXT PRIMARY KEY,
    table_namespace TEXT DEFAULT 'default',
    table_name      TEXT,
    table_path      TEXT,
    table_schema    TEXT,
    properties      TEXT,
    partitions      TEXT,
    domain          TEXT DEFAULT 'public'
);
CREATE INDEX IF NOT EXISTS table_info_name_index ON table_info (table_namespace, table_name);
CREATE INDEX IF NOT EXISTS table_info_path_index ON table_info (table_path);
CREATE TABLE IF NOT EXISTS table_name_id (
    table_name      TEXT,
    table_id        TEXT,
    table_namespace TEXT DEFAULT 'default',
    domain          TEXT DEFAULT 'public',
    PRIMARY KEY (table_name, table_namespace)
);
CREATE TABLE IF NOT EXISTS table_path_id (
    table_path      TEXT PRIMARY KEY,
    table_id        TEXT,
    table_namespace TEXT DEFAULT 'default',
    domain          TEXT DEFAULT 'public'
);
CREATE TABLE IF NOT EXISTS data_commit_info (
    table_id       TEXT,
    partition_desc TEXT,
    commit_id      TEXT,
    file_ops       TEXT,          -- JSON list of data_file_op
    commit_op      TEXT,
    committed      INTEGER DEFAULT 0,
    timestamp      INTEGER,
    domain         TEXT DEFAULT 'public',
    PRIMARY KEY (table_id, partition_desc, commit_id)
);
CREATE TABLE IF NOT EXISTS partition_info (
    table_id       TEXT,
    partition_desc TEXT,
    version        INTEGER,
    commit_op      TEXT,
    timestamp      INTEGER,
    snapshot       TEXT,          -- JSON list of commit UUIDs
    expression     TEXT,
    domain         TEXT DEFAULT 'public',
    PRIMARY KEY (table_id, partition_desc, version)
);
CREATE INDEX IF NOT EXISTS partition_info_timestamp ON partition_info (timestamp);
CREATE TABLE IF NOT EXISTS global_config (
    key   TEXT PRIMARY KEY,
    value TEXT
);
CREATE TABLE IF NOT EXISTS discard_compressed_file_info (
    file_path      TEXT PRIMARY KEY,
    table_path     TEXT,
    partition_desc TEXT,
    timestamp      INTEGER,
    t_date         TEXT
);
"""


def _now_ms() -> int:
    return int(time.time() * 1000)


class SqliteMetaStore:
    """Single-node metadata store on SQLite (WAL, multi-process safe)."""

    def __init__(self, path: Optional[str] = None):
        if path is None:
            path = os.environ.get("LAKESOUL_META_DB", "")
        if not path:
            path = os.path.join(os.getcwd(), ".lakesoul", "meta.db")
        if path != ":memory:":
            os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        self.path = path
        self._local = threading.local()
        conn = self._conn()
        with conn:
            conn.executescript(_SQLITE_SCHEMA)
            conn.execute(
                "INSERT OR IGNORE INTO namespace(namespace, properties, comment, domain)"
                " VALUES ('default', '{}', '', 'public')"
            )

    # -- connection management ---------------------------------------- #

    def _conn(self) -> sqlite3.Connection:
        conn = getattr(self._local, "conn", None)
        if conn is None:
            conn = sqlite3.connect(self.path, timeout=60.0)
            # busy_timeout FIRST: switching journal_mode takes a write lock
            # and races with other ranks opening the same store
            conn.execute("PRAGMA busy_timeout=60000")
            for attempt in range(20):
                try:
                    conn.execute("PRAGMA journal_mode=WAL")
                    break
                except sqlite3.OperationalError:
                    import random
                    import time as _time

                    _time.sleep(random.uniform(0.01, 0.1))
            conn.execute("PRAGMA synchronous=NORMAL")
            self._local.conn = conn
        return conn

    def close(self) -> None:
        conn = getattr(self._local, "conn", None)
        if conn is not None:
            conn.close()
            self._local.conn = None

    # -- namespace ----------------------------------------------------- #

    def insert_namespace(self, ns: Namespace) -> None:
        with self._conn() as c:
            c.execute(
                "INSERT OR REPLACE INTO namespace VALUES (?,?,?,?)",
                (ns.namespace, ns.properties, ns.comment, ns.domain),
            )

    def get_namespace(self, namespace: str) -> Optional[Namespace]:
        cur = self._conn().execute(
            "SELECT namespace, properties, comment, domain FROM namespace WHERE namespace=?",
            (namespace,),
        )
        row = cur.fetchone()
        return Namespace(*row) if row else None

    def list_namespaces(self) -> List[str]:
        cur = self._conn().execute("SELECT namespace FROM namespace ORDER BY namespace")
        return [r[0] for r in cur.fetchall()]

    def delete_namespace(self, namespace: str) -> None:
        with self._conn() as c:
            c.execute("DELETE FROM namespace WHERE namespace=?", (namespace,))

    # -- table_info + ids ---------------------------------------------- #

    def create_table(self, info: TableInfo) -> None:
        """Insert table_info + table_name_id + table_path_id atomically."""
        conn = self._conn()
        with conn:
            conn.execute(
                "INSERT INTO table_info VALUES (?,?,?,?,?,?,?,?)",
                (
                    info.table_id,
                    info.table_namespace,
                    info.table_name,
                    info.table_path,
                    info.table_schema,
                    info.properties,
                    info.partitions,
                    info.domain,
                ),
            )
            if info.table_name:
                conn.execute(
                    "INSERT INTO table_name_id VALUES (?,?,?,?)",
                    (info.table_name, info.table_id, info.table_namespace, info.domain),
                )
            if info.table_path:
                conn.execute(
                    "INSERT INTO table_path_id VALUES (?,?,?,?)",
                    (info.table_path, info.table_id, info.table_namespace, info.domain),
                )

    def update_table_schema(self, table_id: str, schema_json: str) -> None:
        with self._conn() as c:
            c.execute(
                "UPDATE table_info SET table_schema=? WHERE table_id=?",
                (schema_json, table_id),
            )

    def update_table_properties(self, table_id: str, properties: str) -> None:
        with self._conn() as c:
            c.execute(
                "UPDATE table_info SET properties=? WHERE table_id=?",
                (properties, table_id),
            )

    def _row_to_table_info(self, row) -> TableInfo:
        return TableInfo(
            table_id=row[0],
            table_namespace=row[1],
            table_name=row[2],
            table_path=row[3],
            table_schema=row[4],
            properties=row[5],
            partitions=row[6],
            domain=row[7],
        )

    def get_table_info_by_id(self, table_id: str) -> Optional[TableInfo]:
        cur = self._conn().execute(
            "SELECT * FROM table_info WHERE table_id=?", (table_id,)
        )
        row = cur.fetchone()
        return self._row_to_table_info(row) if row else None

    def get_table_info_by_name(
        self, table_name: str, namespace: str = "default"
    ) -> Optional[TableInfo]:
        cur = self._conn().execute(
            "SELECT * FROM table_info WHERE table_name=? AND table_namespace=?",
            (table_name, namespace),
        )
        row = cur.fetchone()
        return self._row_to_table_info(row) if row else None

    def get_table_info_by_path(self, table_path: str) -> Optional[TableInfo]:
        cur = self._conn().execute(
            "SELECT * FROM table_info WHERE table_path=?", (table_path,)
        )
        row = cur.fetchone()
        return self._row_to_table_info(row) if row else None

    def list_tables(self, namespace: str = "default") -> List[TableInfo]:
        cur = self._conn().execute(
            "SELECT * FROM table_info WHERE table_namespace=? ORDER BY table_name",
            (namespace,),
        )
        return [self._row_to_table_info(r) for r in cur.fetchall()]

    def drop_table(self, table_id: str) -> None:
        conn = self._conn()
        with conn:
            conn.execute("DELETE FROM table_name_id WHERE table_id=?", (table_id,))
            conn.execute("DELETE FROM table_path_id WHERE table_id=?", (table_id,))
            conn.execute("DELETE FROM partition_info WHERE table_id=?", (table_id,))
            conn.execute("DELETE FROM data_commit_info WHERE table_id=?", (table_id,))
            conn.execute("DELETE FROM table_info WHERE table_id=?", (table_id,))

    # -- data_commit_info ---------------------------------------------- #

    def insert_data_commit_info(self, dci: DataCommitInfo) -> None:
        with self._conn() as c:
            c.execute(
                "INSERT INTO data_commit_info VALUES (?,?,?,?,?,?,?,?)",
                (
                    dci.table_id,
                    dci.partition_desc,
                    dci.commit_id,
                    json.dumps([op.to_json() for op in dci.file_ops]),
                    dci.commit_op.name,
                    1 if dci.committed else 0,
                    dci.timestamp,
                    dci.domain,
                ),
            )

    def set_commit_committed(
        self, table_id: str, partition_desc: str, commit_id: str
    ) -> None:
        with self._conn() as c:
            c.execute(
                "UPDATE data_commit_info SET committed=1"
                " WHERE table_id=? AND partition_desc=? AND commit_id=?",
                (table_id, partition_desc, commit_id),
            )

    def _row_to_dci(self, row) -> DataCommitInfo:
        return DataCommitInfo(
            table_id=row[0],
            partition_desc=row[1],
            commit_id=row[2],
            file_ops=[DataFileOp.from_json(d) for d in json.loads(row[3] or "[]")],
            commit_op=CommitOp.from_name(row[4]),
            committed=bool(row[5]),
            timestamp=row[6],
            domain=row[7],
        )

    def get_data_commit_info(
        self, table_id: str, partition_desc: str, commit_id: str
    ) -> Optional[DataCommitInfo]:
        cur = self._conn().execute(
            "SELECT * FROM data_commit_info"
            " WHERE table_id=? AND partition_desc=? AND commit_id=?",
            (table_id, partition_desc, commit_id),
        )
        row = cur.fetchone()
        return self._row_to_dci(row) if row else None

    def get_data_commits(
        self, table_id: str, partition_desc: str, commit_ids: Sequence[str]
    ) -> List[DataCommitInfo]:
        """Fetch commits preserving the order of commit_ids (snapshot order)."""
        if not commit_ids:
            return []
        qmarks = ",".join("?" for _ in commit_ids)
        cur = self._conn().execute(
            f"SELECT * FROM data_commit_info WHERE table_id=? AND partition_desc=?"
            f" AND commit_id IN ({qmarks})",
            (table_id, partition_desc, *commit_ids),
        )
        by_id = {r[2]: self._row_to_dci(r) for r in cur.fetchall()}
        return [by_id[cid] for cid in commit_ids if cid in by_id]

    def delete_data_commit_info(
        self, table_id: str, partition_desc: str, commit_id: str
    ) -> None:
        with self._conn() as c:
            c.execute(
                "DELETE FROM data_commit_info"
                " WHERE table_id=? AND partition_desc=? AND commit_id=?",
                (table_id, partition_desc, commit_id),
            )

    # -- partition_info (MVCC) ----------------------------------------- #

    def _row_to_partition_info(self, row) -> PartitionInfo:
        return PartitionInfo(
            table_id=row[0],
            partition_desc=row[1],
            version=row[2],
            commit_op=CommitOp.from_name(row[3]) if row[3] else CommitOp.AppendCommit,
            timestamp=row[4],
            snapshot=json.loads(row[5] or "[]"),
            expression=row[6] or "",
            domain=row[7],
        )

    def get_latest_partition_info(
        self, table_id: str, partition_desc: str
    ) -> Optional[PartitionInfo]:
        cur = self._conn().execute(
            "SELECT * FROM partition_info WHERE table_id=? AND partition_desc=?"
            " ORDER BY version DESC LIMIT 1",
            (table_id, partition_desc),
        )
        row = cur.fetchone()
        return self._row_to_partition_info(row) if row else None

    def get_partition_info_by_version(
        self, table_id: str, partition_desc: str, version: int
    ) -> Optional[PartitionInfo]:
        cur = self._conn().execute(
            "SELECT * FROM partition_info WHERE table_id=? AND partition_desc=? AND version=?",
            (table_id, partition_desc, version),
        )
        row = cur.fetchone()
        return self._row_to_partition_info(row) if row else None

    def get_latest_partition_info_before(
        self, table_id: str, partition_desc: str, ts_ms: int
    ) -> Optional[PartitionInfo]:
        """Time-travel: latest version with timestamp <= ts_ms."""
        cur = self._conn().execute(
            "SELECT * FROM partition_info WHERE table_id=? AND partition_desc=?"
            " AND timestamp<=? ORDER BY version DESC LIMIT 1",
            (table_id, partition_desc, ts_ms),
        )
        row = cur.fetchone()
        return self._row_to_partition_info(row) if row else None

    def get_partition_versions_in_range(
        self,
        table_id: str,
        partition_desc: str,
        start_version: int,
        end_version: int,
    ) -> List[PartitionInfo]:
        cur = self._conn().execute(
            "SELECT * FROM partition_info WHERE table_id=? AND partition_desc=?"
            " AND version>=? AND version<=? ORDER BY version",
            (table_id, partition_desc, start_version, end_version),
        )
        return [self._row_to_partition_info(r) for r in cur.fetchall()]

    def get_all_partition_desc(self, table_id: str) -> List[str]:
        cur = self._conn().execute(
            "SELECT DISTINCT partition_desc FROM partition_info WHERE table_id=?"
            " ORDER BY partition_desc",
            (table_id,),
        )
        return [r[0] for r in cur.fetchall()]

    def get_all_partition_info(self, table_id: str) -> List[PartitionInfo]:
        """Latest version per partition_desc."""
        cur = self._conn().execute(
            "SELECT p.* FROM partition_info p JOIN ("
            "  SELECT partition_desc, MAX(version) AS v FROM partition_info"
            "  WHERE table_id=? GROUP BY partition_desc"
            ") m ON p.partition_desc=m.partition_desc AND p.version=m.v"
            " WHERE p.table_id=?",
            (table_id, table_id),
        )
        return [self._row_to_partition_info(r) for r in cur.fetchall()]

    def transaction_insert_partition_info(
        self, partitions: List[PartitionInfo]
    ) -> None:
        """Atomically insert new partition versions; PK conflict raises
        CommitConflictError (the MVCC CAS — reference:
        lakesoul-metadata/src/lib.rs:212-216 transaction_insert_partition_info)."""
        conn = self._conn()
        try:
            with conn:
                for p in partitions:
                    conn.execute(
                        "INSERT INTO partition_info VALUES (?,?,?,?,?,?,?,?)",
                        (
                            p.table_id,
                            p.partition_desc,
                            p.version,
                            p.commit_op.name,
                            p.timestamp or _now_ms(),
                            json.dumps(p.snapshot),
                            p.expression,
                            p.domain,
                        ),
                    )
                # flip two-phase flags inside the same transaction, as the
                # reference does (transaction_insert_partition_info updates
                # data_commit_info.committed in the same PG transaction)
                for p in partitions:
                    for cid in p.snapshot:
                        conn.execute(
                            "UPDATE data_commit_info SET committed=1"
                            " WHERE table_id=? AND partition_desc=? AND commit_id=?",
                            (p.table_id, p.partition_desc, cid),
                        )
        except sqlite3.IntegrityError as e:
            raise CommitConflictError(str(e)) from e

    def delete_partition_versions_since(
        self, table_id: str, partition_desc: str, version: int
    ) -> None:
        """Rollback helper: drop versions >= version."""
        with self._conn() as c:
            c.execute(
                "DELETE FROM partition_info WHERE table_id=? AND partition_desc=?"
                " AND version>=?",
                (table_id, partition_desc, version),
            )

    def vacuum_partition_versions(
        self, table_id: str, partition_desc: str, cutoff: int,
        stale_commit_ids: List[str],
    ) -> None:
        """Vacuum: atomically drop partition versions < cutoff and their
        no-longer-referenced commit infos in ONE transaction (a reader or
        concurrent committer never observes an empty version history —
        ROADMAP hygiene item)."""
        with self._conn() as c:
            c.execute(
                "DELETE FROM partition_info WHERE table_id=? AND"
                " partition_desc=? AND version<?",
                (table_id, partition_desc, cutoff),
            )
            for cid in stale_commit_ids:
                c.execute(
                    "DELETE FROM data_commit_info WHERE table_id=? AND"
                    " partition_desc=? AND commit_id=?",
                    (table_id, partition_desc, cid),
                )

    # -- global config -------------------------------------------------- #

    def set_global_config(self, key: str, value: str) -> None:
        with self._conn() as c:
            c.execute(
                "INSERT OR REPLACE INTO global_config VALUES (?,?)", (key, value)
            )

    def get_global_config(self, key: str) -> Optional[str]:
        cur = self._conn().execute(
            "SELECT value FROM global_config WHERE key=?", (key,)
        )
        row = cur.fetchone()
        return row[0] if row else None

    # -- discarded files (compaction cleanup) --------------------------- #

    def insert_discard_file(
        self, file_path: str, table_path: str, partition_desc: str
    ) -> None:
        with self._conn() as c:
            c.execute(
                "INSERT OR REPLACE INTO discard_compressed_file_info VALUES (?,?,?,?,?)",
                (file_path, table_path, partition_desc, _now_ms(), ""),
            )

    def list_discard_files(self, table_path: Optional[str] = None) -> List[str]:
        if table_path is None:
            cur = self._conn().execute(
                "SELECT file_path FROM discard_compressed_file_info"
            )
        else:
            cur = self._conn().execute(
                "SELECT file_path FROM discard_compressed_file_info WHERE table_path=?",
                (table_path,),
            )
        return [r[0] for r in cur.fetchall()]

    def delete_discard_file(self, file_path: str) -> None:
        with self._conn() as c:
            c.execute(
                "DELETE FROM discard_compressed_file_info WHERE file_path=?",
                (file_path,),
            )

    # -- test helper ---------------------------------------------------- #

    def clean_meta_for_test(self) -> None:
        conn = self._conn()
        with conn:
            for t in (
                "namespace",
                "table_info",
                "table_name_id",
                "table_path_id",
                "data_commit_info",
                "partition_info",
                "global_config",
                "discard_compressed_file_info",
            ):
                conn.execute(f"DELETE FROM {t}")
            conn.execute(
                "INSERT INTO namespace(namespace, properties, comment, domain)"
                " VALUES ('default', '{}', '', 'public')"
            )


def open_meta_store(uri: Optional[str] = None) -> SqliteMetaStore:
    """Open the configured metadata store.

    ``LAKESOUL_PG_URL`` selects the PostgreSQL backend when a driver is
    available; otherwise SQLite at ``LAKESOUL_META_DB`` (or
    ``./.lakesoul/meta.db``).
    """
    pg_url = os.environ.get("LAKESOUL_PG_URL", "")
    if pg_url:
        try:
            from .pg_store import PostgresMetaStore  # type: ignore

            return PostgresMetaStore(pg_url)  # pragma: no cover
        except ImportError:
            raise RuntimeError(
                "LAKESOUL_PG_URL set but no postgres driver (psycopg) available"
            )
    return SqliteMetaStore(uri)
