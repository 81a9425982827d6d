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
        if psycopg is None:  # pragma: no cover
            raise ImportError(
                "PostgresMetaStore needs psycopg/psycopg2 (not available in "
                "the offline build image); unset LAKESOUL_PG_URL to use SQLite"
            )
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

    # -- namespace ------------------------------------------------------ #

    def get_namespace(self, namespace: str):
        rows = self._exec(
            "SELECT namespace, properties::text, comment, domain FROM namespace"
            " WHERE namespace=%s", (namespace,))
        return Namespace(*rows[0]) if rows else None

    def delete_namespace(self, namespace: str) -> None:
        self._exec("DELETE FROM namespace WHERE namespace=%s", (namespace,))

    def close(self) -> None:
        self.conn.close()

    # -- table_info ------------------------------------------------------ #

    _TI_COLS = ("table_id, table_namespace, table_name, table_path,"
                " table_schema, properties::text, partitions, domain")

    def _row_to_table_info(self, row) -> TableInfo:
        return TableInfo(*row)

    def get_table_info_by_id(self, table_id: str):
        rows = self._exec(
            f"SELECT {self._TI_COLS} FROM table_info WHERE table_id=%s",
            (table_id,))
        return self._row_to_table_info(rows[0]) if rows else None

    def get_table_info_by_path(self, table_path: str):
        rows = self._exec(
            f"SELECT {self._TI_COLS} FROM table_info WHERE table_path=%s",
            (table_path,))
        return self._row_to_table_info(rows[0]) if rows else None

    def list_tables(self, namespace: str = "default") -> List[TableInfo]:
        rows = self._exec(
            f"SELECT {self._TI_COLS} FROM table_info WHERE table_namespace=%s"
            " ORDER BY table_name", (namespace,))
        return [self._row_to_table_info(r) for r in rows]

    def update_table_schema(self, table_id: str, schema_json: str) -> None:
        self._exec("UPDATE table_info SET table_schema=%s WHERE table_id=%s",
                   (schema_json, table_id))

    def update_table_properties(self, table_id: str, properties: str) -> None:
        self._exec("UPDATE table_info SET properties=%s::json WHERE table_id=%s",
                   (properties, table_id))

    def drop_table(self, table_id: str) -> None:
        with self.conn.transaction():
            with self.conn.cursor() as cur:
                for t in ("table_name_id", "table_path_id", "partition_info",
                          "data_commit_info", "table_info"):
                    cur.execute(f"DELETE FROM {t} WHERE table_id=%s", (table_id,))

    # -- data_commit_info ------------------------------------------------ #

    def _row_to_dci(self, row) -> DataCommitInfo:
        import json as _json

        ops = [
            DataFileOp(d["path"], FileOp[d["file_op"]], d.get("size", 0),
                       d.get("file_exist_cols", ""))
            for d in _json.loads(row[3] or "[]")
        ]
        return DataCommitInfo(
            table_id=row[0], partition_desc=row[1], commit_id=row[2],
            file_ops=ops, commit_op=CommitOp.from_name(row[4]),
            committed=bool(row[5]), timestamp=row[6], domain=row[7])

    _DCI_COLS = ("table_id, partition_desc, commit_id,"
                 " array_to_json(file_ops)::text, commit_op, committed,"
                 " timestamp, domain")

    def get_data_commit_info(self, table_id, partition_desc, commit_id):
        rows = self._exec(
            f"SELECT {self._DCI_COLS} FROM data_commit_info WHERE table_id=%s"
            " AND partition_desc=%s AND commit_id=%s",
            (table_id, partition_desc, commit_id))
        return self._row_to_dci(rows[0]) if rows else None

    def get_data_commits(self, table_id, partition_desc, commit_ids):
        if not commit_ids:
            return []
        rows = self._exec(
            f"SELECT {self._DCI_COLS} FROM data_commit_info WHERE table_id=%s"
            " AND partition_desc=%s AND commit_id=ANY(%s::uuid[])",
            (table_id, partition_desc, list(commit_ids)))
        by_id = {r[2]: self._row_to_dci(r) for r in rows}
        return [by_id[cid] for cid in commit_ids if cid in by_id]

    def set_commit_committed(self, table_id, partition_desc, commit_id) -> None:
        self._exec(
            "UPDATE data_commit_info SET committed=true WHERE table_id=%s"
            " AND partition_desc=%s AND commit_id=%s",
            (table_id, partition_desc, commit_id))

    def delete_data_commit_info(self, table_id, partition_desc, commit_id) -> None:
        self._exec(
            "DELETE FROM data_commit_info WHERE table_id=%s"
            " AND partition_desc=%s AND commit_id=%s",
            (table_id, partition_desc, commit_id))

    # -- partition_info --------------------------------------------------- #

    _PI_COLS = ("table_id, partition_desc, version, commit_op, timestamp,"
                " array_to_json(snapshot)::text, expression, domain")

    def _row_to_partition_info(self, row) -> PartitionInfo:
        import json as _json

        return PartitionInfo(
            table_id=row[0], partition_desc=row[1], version=row[2],
            commit_op=CommitOp.from_name(row[3]) if row[3] else CommitOp.AppendCommit,
            timestamp=row[4], snapshot=_json.loads(row[5] or "[]"),
            expression=row[6] or "", domain=row[7])

    def get_partition_info_by_version(self, table_id, partition_desc, version):
        rows = self._exec(
            f"SELECT {self._PI_COLS} FROM partition_info WHERE table_id=%s"
            " AND partition_desc=%s AND version=%s",
            (table_id, partition_desc, version))
        return self._row_to_partition_info(rows[0]) if rows else None

    def get_latest_partition_info_before(self, table_id, partition_desc, ts_ms):
        rows = self._exec(
            f"SELECT {self._PI_COLS} FROM partition_info WHERE table_id=%s"
            " AND partition_desc=%s AND timestamp<=%s ORDER BY version DESC LIMIT 1",
            (table_id, partition_desc, ts_ms))
        return self._row_to_partition_info(rows[0]) if rows else None

    def get_partition_versions_in_range(self, table_id, partition_desc,
                                        start_version, end_version):
        rows = self._exec(
            f"SELECT {self._PI_COLS} FROM partition_info WHERE table_id=%s"
            " AND partition_desc=%s AND version>=%s AND version<=%s"
            " ORDER BY version",
            (table_id, partition_desc, start_version, end_version))
        return [self._row_to_partition_info(r) for r in rows]

    def get_all_partition_desc(self, table_id) -> List[str]:
        rows = self._exec(
            "SELECT DISTINCT partition_desc FROM partition_info"
            " WHERE table_id=%s ORDER BY partition_desc", (table_id,))
        return [r[0] for r in rows]

    def get_all_partition_info(self, table_id) -> List[PartitionInfo]:
        rows = self._exec(
            f"SELECT {self._PI_COLS.replace('table_id', 'p.table_id')}"
            " FROM partition_info p JOIN ("
            "  SELECT partition_desc, MAX(version) AS v FROM partition_info"
            "  WHERE table_id=%s GROUP BY partition_desc"
            ") m ON p.partition_desc=m.partition_desc AND p.version=m.v"
            " WHERE p.table_id=%s", (table_id, table_id))
        return [self._row_to_partition_info(r) for r in rows]

    def delete_partition_versions_since(self, table_id, partition_desc,
                                        version) -> None:
        self._exec(
            "DELETE FROM partition_info WHERE table_id=%s AND"
            " partition_desc=%s AND version>=%s",
            (table_id, partition_desc, version))

    # -- global config / discard files ------------------------------------ #

    def set_global_config(self, key: str, value: str) -> None:
        self._exec(
            "INSERT INTO global_config (key, value) VALUES (%s,%s)"
            " ON CONFLICT (key) DO UPDATE SET value=EXCLUDED.value",
            (key, value))

    def get_global_config(self, key: str):
        rows = self._exec("SELECT value FROM global_config WHERE key=%s", (key,))
        return rows[0][0] if rows else None

    def insert_discard_file(self, file_path, table_path, partition_desc) -> None:
        self._exec(
            "INSERT INTO discard_compressed_file_info"
            " (file_path, table_path, partition_desc, timestamp, t_date)"
            " VALUES (%s,%s,%s,(extract(epoch from now())*1000)::bigint,"
            " to_char(now(), 'YYYY-MM-DD')) ON CONFLICT (file_path) DO NOTHING",
            (file_path, table_path, partition_desc))

    def list_discard_files(self, table_path=None) -> List[str]:
        if table_path:
            rows = self._exec(
                "SELECT file_path FROM discard_compressed_file_info"
                " WHERE table_path=%s", (table_path,))
        else:
            rows = self._exec(
                "SELECT file_path FROM discard_compressed_file_info")
        return [r[0] for r in rows]

    def delete_discard_file(self, file_path: str) -> None:
        self._exec(
            "DELETE FROM discard_compressed_file_info WHERE file_path=%s",
            (file_path,))

    def clean_meta_for_test(self) -> None:
        for t in ("partition_info", "data_commit_info", "table_name_id",
                  "table_path_id", "table_info",
                  "discard_compressed_file_info"):
            self._exec(f"DELETE FROM {t}")

    def listen_compaction(self):
        """Yield compaction notifications from the reference's PG trigger
        channel (meta_init.sql:121-136 pg_notify)."""
        self._exec("LISTEN lakesoul_compaction_notify")
        gen = self.conn.notifies()
        for note in gen:
            yield json.loads(note.payload)
