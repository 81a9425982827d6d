"""PostgreSQL metadata store backend.

Speaks to the reference's actual catalog schema (``script/meta_init.sql``
— apply it to the database first): UUID[] snapshots, data_file_op[]
composites, the partition_insert trigger, pg_notify compaction channel.
Same DAO surface as SqliteMetaStore, so MetaClient works unchanged; the
MVCC CAS maps to the partition_info primary-key insert conflict exactly
as in the reference (DBManager.java:509 / metadata_client.rs:498).

Requires a psycopg driver (``psycopg`` or ``psycopg2``) — not installed
in the offline build image, so this backend is import-gated; enable with
``LAKESOUL_PG_URL=postgresql://user:pass@host/db``.
"""

from __future__ import annotations

import json
from typing import List, Optional, Sequence

try:
    import psycopg  # type: ignore

    _DRIVER = "psycopg"
except ImportError:  # pragma: no cover
    try:
        import psycopg2 as psycopg  # type: ignore

        _DRIVER = "psycopg2"
    except ImportError:
        psycopg = None
        _DRIVER = None

if psycopg is None:  # pragma: no cover
    raise ImportError(
        "PostgresMetaStore needs psycopg/psycopg2 (not available in the "
        "offline build image); unset LAKESOUL_PG_URL to use SQLite"
    )

from .entities import (  # noqa: E402
    CommitOp,
    DataCommitInfo,
    DataFileOp,
    FileOp,
    Namespace,
    PartitionInfo,
    TableInfo,
)
from .store import CommitConflictError  # noqa: E402


class PostgresMetaStore:  # pragma: no cover — needs a live PG server
    """Mirror of SqliteMetaStore over the reference PG schema."""

    def __init__(self, url: str):
        self.url = url
        self.conn = psycopg.connect(url)
        self.conn.autocommit = True
        self.path = url  # parity with SqliteMetaStore.path

    # -- helpers -------------------------------------------------------- #

    def _exec(self, sql: str, params=()):
        with self.conn.cursor() as cur:
            cur.execute(sql, params)
            if cur.description:
                return cur.fetchall()
            return None

    @staticmethod
    def _file_ops_to_pg(ops: Sequence[DataFileOp]) -> list:
        return [(o.path, o.file_op.text, o.size, o.file_exist_cols) for o in ops]

    # -- namespace ------------------------------------------------------ #

    def insert_namespace(self, ns: Namespace) -> None:
        self._exec(
            "INSERT INTO namespace VALUES (%s,%s,%s,%s) ON CONFLICT DO NOTHING",
            (ns.namespace, ns.properties, ns.comment, ns.domain),
        )

    def list_namespaces(self) -> List[str]:
        return [r[0] for r in self._exec("SELECT namespace FROM namespace ORDER BY 1")]

    # -- tables --------------------------------------------------------- #

    def create_table(self, info: TableInfo) -> None:
        with self.conn.cursor() as cur:
            cur.execute(
                "INSERT INTO table_info(table_id, table_namespace, table_name,"
                " table_path, table_schema, properties, partitions, domain)"
                " VALUES (%s,%s,%s,%s,%s,%s,%s,%s)",
                (
                    info.table_id, info.table_namespace, info.table_name,
                    info.table_path, info.table_schema, info.properties,
                    info.partitions, info.domain,
                ),
            )
            if info.table_name:
                cur.execute(
                    "INSERT INTO table_name_id VALUES (%s,%s,%s,%s)",
                    (info.table_name, info.table_id, info.table_namespace, info.domain),
                )
            if info.table_path:
                cur.execute(
                    "INSERT INTO table_path_id VALUES (%s,%s,%s,%s)",
                    (info.table_path, info.table_id, info.table_namespace, info.domain),
                )

    def get_table_info_by_name(self, name: str, namespace: str = "default"):
        rows = self._exec(
            "SELECT table_id, table_namespace, table_name, table_path,"
            " table_schema, properties, partitions, domain FROM table_info"
            " WHERE table_name=%s AND table_namespace=%s",
            (name, namespace),
        )
        return TableInfo(*rows[0]) if rows else None

    # -- partition_info MVCC -------------------------------------------- #

    def get_latest_partition_info(self, table_id: str, desc: str):
        rows = self._exec(
            "SELECT table_id, partition_desc, version, commit_op, timestamp,"
            " snapshot, expression, domain FROM partition_info"
            " WHERE table_id=%s AND partition_desc=%s ORDER BY version DESC LIMIT 1",
            (table_id, desc),
        )
        if not rows:
            return None
        r = rows[0]
        return PartitionInfo(
            table_id=r[0], partition_desc=r[1], version=r[2],
            commit_op=CommitOp.from_name(r[3]), timestamp=r[4],
            snapshot=[str(u) for u in (r[5] or [])], expression=r[6] or "",
            domain=r[7],
        )

    def transaction_insert_partition_info(self, partitions: List[PartitionInfo]) -> None:
        try:
            with self.conn.transaction():  # psycopg3
                with self.conn.cursor() as cur:
                    for p in partitions:
                        cur.execute(
                            "INSERT INTO partition_info(table_id, partition_desc,"
                            " version, commit_op, snapshot, expression, domain)"
                            " VALUES (%s,%s,%s,%s,%s::uuid[],%s,%s)",
                            (
                                p.table_id, p.partition_desc, p.version,
                                p.commit_op.name, p.snapshot, p.expression, p.domain,
                            ),
                        )
                        for cid in p.snapshot:
                            cur.execute(
                                "UPDATE data_commit_info SET committed=true"
                                " WHERE table_id=%s AND partition_desc=%s AND commit_id=%s",
                                (p.table_id, p.partition_desc, cid),
                            )
        except Exception as e:  # unique violation = MVCC CAS failure
            raise CommitConflictError(str(e)) from e

    def vacuum_partition_versions(self, table_id, partition_desc, cutoff,
                                  stale_commit_ids) -> None:
        """Atomic vacuum (same contract as SqliteMetaStore)."""
        with self.conn.transaction():
            with self.conn.cursor() as cur:
                cur.execute(
                    "DELETE FROM partition_info WHERE table_id=%s AND"
                    " partition_desc=%s AND version<%s",
                    (table_id, partition_desc, cutoff),
                )
                for cid in stale_commit_ids:
                    cur.execute(
                        "DELETE FROM data_commit_info WHERE table_id=%s AND"
                        " partition_desc=%s AND commit_id=%s",
                        (table_id, partition_desc, cid),
                    )

    def insert_data_commit_info(self, dci: DataCommitInfo) -> None:
        self._exec(
            "INSERT INTO data_commit_info(table_id, partition_desc, commit_id,"
            " file_ops, commit_op, committed, timestamp, domain)"
            " VALUES (%s,%s,%s,%s::data_file_op[],%s,%s,%s,%s)",
            (
                dci.table_id, dci.partition_desc, dci.commit_id,
                self._file_ops_to_pg(dci.file_ops), dci.commit_op.name,
                dci.committed, dci.timestamp, dci.domain,
            ),
        )

    def listen_compaction(self):
        """Yield compaction notifications from the reference's PG trigger
        channel (meta_init.sql:121-136 pg_notify)."""
        self._exec("LISTEN lakesoul_compaction_notify")
        gen = self.conn.notifies()
        for note in gen:
            yield json.loads(note.payload)
