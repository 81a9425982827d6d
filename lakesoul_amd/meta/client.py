"""High-level metadata client.

MI355X-native analog of the reference's ``MetaDataClient``
(``rust/lakesoul-metadata/src/metadata_client.rs``): the MVCC
``commit_data`` protocol (metadata_client.rs:498-663), the two-phase
``commit_data_commit_info`` (metadata_client.rs:690), snapshot /
incremental queries (metadata_client.rs:1052-1126) and the compaction
trigger rule (script/meta_init.sql:102-150).
"""

from __future__ import annotations

import random
import time
from typing import Dict, List, Optional, Sequence

from .entities import (
    CommitOp,
    DataCommitInfo,
    DataFileOp,
    MetaInfo,
    Namespace,
    PartitionInfo,
    TableInfo,
)
from .store import CommitConflictError, SqliteMetaStore, open_meta_store

DEFAULT_MAX_RETRY = 25


class ConcurrentReplaceError(Exception):
    """A Compaction/Update commit raced another snapshot REPLACE: the
    compacted output is based on a snapshot that no longer exists. Not
    retryable at the commit layer — the caller must redo its work from
    the new snapshot."""


class MetaClient:
    def __init__(self, store: Optional[SqliteMetaStore] = None, max_retry: int = DEFAULT_MAX_RETRY):
        self.store = store if store is not None else open_meta_store()
        self.max_retry = max_retry

    # ------------------------------------------------------------------ #
    # namespaces / tables
    # ------------------------------------------------------------------ #

    def create_namespace(self, namespace: str, properties: str = "{}", comment: str = "") -> None:
        self.store.insert_namespace(Namespace(namespace, properties, comment))

    def list_namespaces(self) -> List[str]:
        return self.store.list_namespaces()

    def create_table(self, info: TableInfo) -> None:
        self.store.create_table(info)

    def get_table_info_by_name(self, name: str, namespace: str = "default") -> Optional[TableInfo]:
        return self.store.get_table_info_by_name(name, namespace)

    def get_table_info_by_path(self, path: str) -> Optional[TableInfo]:
        return self.store.get_table_info_by_path(path)

    def get_table_info_by_id(self, table_id: str) -> Optional[TableInfo]:
        return self.store.get_table_info_by_id(table_id)

    def drop_table(self, table_id: str) -> None:
        self.store.drop_table(table_id)

    def list_tables(self, namespace: str = "default") -> List[TableInfo]:
        return self.store.list_tables(namespace)

    # ------------------------------------------------------------------ #
    # two-phase commit
    # ------------------------------------------------------------------ #

    def commit_data_commit_info(self, dci: DataCommitInfo) -> None:
        """Phase 1: persist the (uncommitted) DataCommitInfo, then run
        commit_data to flip the partition version (reference:
        metadata_client.rs:690 + DBManager.commitDataCommitInfo)."""
        self.store.insert_data_commit_info(dci)
        meta_info = MetaInfo(
            table_info=self.store.get_table_info_by_id(dci.table_id),
            list_partition=[
                PartitionInfo(
                    table_id=dci.table_id,
                    partition_desc=dci.partition_desc,
                    snapshot=[dci.commit_id],
                    commit_op=dci.commit_op,
                )
            ],
        )
        self.commit_data(meta_info, dci.commit_op)

    # ------------------------------------------------------------------ #
    # commit_data: the MVCC protocol
    # ------------------------------------------------------------------ #

    def commit_data(self, meta_info: MetaInfo, commit_op: CommitOp) -> None:
        """Advance partition versions atomically, retrying on CAS conflicts
        (reference semantics, metadata_client.rs:498-663):

        - Append/Merge: snapshot := current ++ new commits, version += 1
        - Compaction/Update: snapshot := new commits (replace), version += 1
        - Delete: snapshot := [], version += 1
        """
        last_err: Optional[Exception] = None
        for attempt in range(self.max_retry):
            try:
                self._commit_data_once(meta_info, commit_op)
                self._maybe_notify_compaction(meta_info, commit_op)
                return
            except CommitConflictError as e:
                last_err = e
                # jittered backoff then re-read current versions
                time.sleep(random.uniform(0.001, 0.01) * (attempt + 1))
        raise CommitConflictError(
            f"commit_data failed after {self.max_retry} retries: {last_err}"
        )

    def _maybe_notify_compaction(self, meta_info: MetaInfo,
                                 commit_op: CommitOp) -> None:
        """Publisher half of the PG trigger (meta_init.sql:102-150): on a
        delta commit that crosses the threshold, fire the compaction
        channel. No-op unless a notify bus is attached."""
        bus = getattr(self, "notify_bus", None)
        if bus is None or commit_op not in (CommitOp.AppendCommit,
                                            CommitOp.MergeCommit):
            return
        from .notify import CompactionEvent

        ti = meta_info.table_info
        for part in meta_info.list_partition:
            if self.compaction_needed(ti.table_id, part.partition_desc):
                cur = self.store.get_latest_partition_info(
                    ti.table_id, part.partition_desc)
                bus.publish(CompactionEvent(
                    ti.table_id, part.partition_desc,
                    cur.version if cur else 0))

    def _commit_data_once(self, meta_info: MetaInfo, commit_op: CommitOp) -> None:
        table_info = meta_info.table_info
        if table_info is None:
            raise ValueError("table info missing")
        domain = table_info.domain

        new_partition_list: List[PartitionInfo] = []
        read_partition_map: Dict[str, PartitionInfo] = {
            p.partition_desc: p for p in meta_info.read_partition_info
        }

        for part in meta_info.list_partition:
            cur = self.store.get_latest_partition_info(
                table_info.table_id, part.partition_desc
            )
            if commit_op in (CommitOp.AppendCommit, CommitOp.MergeCommit):
                if cur is not None:
                    nxt = PartitionInfo(
                        table_id=cur.table_id,
                        partition_desc=cur.partition_desc,
                        version=cur.version + 1,
                        commit_op=commit_op,
                        snapshot=list(cur.snapshot) + list(part.snapshot),
                        expression=part.expression,
                        domain=domain,
                    )
                else:
                    nxt = PartitionInfo(
                        table_id=table_info.table_id,
                        partition_desc=part.partition_desc,
                        version=0,
                        commit_op=commit_op,
                        snapshot=list(part.snapshot),
                        expression=part.expression,
                        domain=domain,
                    )
            elif commit_op in (CommitOp.CompactionCommit, CommitOp.UpdateCommit):
                if cur is None:
                    cur = PartitionInfo(
                        table_id=table_info.table_id,
                        partition_desc=part.partition_desc,
                        version=-1,
                        domain=domain,
                    )
                read_info = read_partition_map.get(part.partition_desc)
                # snapshot replacement. The reference leaves the
                # concurrent-change case TODO (metadata_client.rs:609-620);
                # here the strict rule is: the replacement covers exactly
                # the snapshot the compaction READ. Commits that landed
                # after the read (cur.snapshot extends read.snapshot) are
                # re-appended ON TOP of the compacted commit; if the
                # current snapshot is NOT an extension of the read one
                # (another replace won the race), this compaction is stale
                # and must abort.
                if read_info is None or cur.version < 0 or                         read_info.version == cur.version:
                    snapshot = list(part.snapshot)
                else:
                    rs = list(read_info.snapshot)
                    cs = list(cur.snapshot)
                    if cs[: len(rs)] != rs:
                        raise ConcurrentReplaceError(
                            f"partition {part.partition_desc}: snapshot was "
                            f"replaced concurrently (read v{read_info.version},"
                            f" now v{cur.version}) — redo the compaction"
                        )
                    snapshot = list(part.snapshot) + cs[len(rs):]
                nxt = PartitionInfo(
                    table_id=table_info.table_id,
                    partition_desc=part.partition_desc,
                    version=cur.version + 1,
                    commit_op=commit_op,
                    snapshot=snapshot,
                    expression=part.expression,
                    domain=domain,
                )
            elif commit_op is CommitOp.DeleteCommit:
                if cur is None:
                    continue
                nxt = PartitionInfo(
                    table_id=table_info.table_id,
                    partition_desc=part.partition_desc,
                    version=cur.version + 1,
                    commit_op=commit_op,
                    snapshot=[],
                    expression=part.expression,
                    domain=domain,
                )
            else:  # pragma: no cover
                raise ValueError(f"unknown commit op {commit_op}")
            new_partition_list.append(nxt)

        if new_partition_list:
            self.store.transaction_insert_partition_info(new_partition_list)

    # ------------------------------------------------------------------ #
    # scan-side queries
    # ------------------------------------------------------------------ #

    def get_latest_version(self, table_id: str, partition_desc: str) -> Optional[int]:
        p = self.store.get_latest_partition_info(table_id, partition_desc)
        return p.version if p else None

    def files_for_partition(
        self,
        table_id: str,
        partition_desc: str,
        version: Optional[int] = None,
        timestamp_ms: Optional[int] = None,
    ) -> List[DataFileOp]:
        """Resolve the file list visible at a snapshot.

        Walks the partition's snapshot commit UUIDs in order applying
        add/del file ops (reference: DataFileInfo transfusion,
        rust/lakesoul-metadata/src/transfusion.rs:316-403).
        """
        if timestamp_ms is not None:
            part = self.store.get_latest_partition_info_before(
                table_id, partition_desc, timestamp_ms
            )
        elif version is not None:
            part = self.store.get_partition_info_by_version(
                table_id, partition_desc, version
            )
        else:
            part = self.store.get_latest_partition_info(table_id, partition_desc)
        if part is None:
            return []
        return self._resolve_snapshot_files(table_id, partition_desc, part.snapshot)

    def _resolve_snapshot_files(
        self, table_id: str, partition_desc: str, snapshot: Sequence[str]
    ) -> List[DataFileOp]:
        commits = self.store.get_data_commits(table_id, partition_desc, snapshot)
        out: List[DataFileOp] = []
        seen: Dict[str, int] = {}
        for dci in commits:
            for op in dci.file_ops:
                if op.file_op.text == "add":
                    if op.path in seen:
                        out[seen[op.path]] = op
                    else:
                        seen[op.path] = len(out)
                        out.append(op)
                else:  # del
                    if op.path in seen:
                        idx = seen.pop(op.path)
                        out[idx] = None  # type: ignore
        return [o for o in out if o is not None]

    def incremental_files(
        self,
        table_id: str,
        partition_desc: str,
        start_version: int,
        end_version: int,
    ) -> List[DataFileOp]:
        """Files added in (start_version, end_version] — incremental read
        (reference: metadata_client.rs:1052-1126)."""
        parts = self.store.get_partition_versions_in_range(
            table_id, partition_desc, start_version + 1, end_version
        )
        commit_ids: List[str] = []
        for p in parts:
            if p.commit_op is CommitOp.CompactionCommit:
                # compaction rewrites existing data — not incremental rows
                continue
            for cid in p.snapshot:
                if cid not in commit_ids:
                    commit_ids.append(cid)
        # drop commits already visible at start_version
        base = self.store.get_partition_info_by_version(
            table_id, partition_desc, start_version
        )
        base_ids = set(base.snapshot) if base else set()
        new_ids = [c for c in commit_ids if c not in base_ids]
        return self._resolve_snapshot_files(table_id, partition_desc, new_ids)

    def all_partition_descs(self, table_id: str) -> List[str]:
        return self.store.get_all_partition_desc(table_id)

    def all_partition_info(self, table_id: str) -> List[PartitionInfo]:
        return self.store.get_all_partition_info(table_id)

    # ------------------------------------------------------------------ #
    # compaction trigger (pg_notify analog)
    # ------------------------------------------------------------------ #

    def compaction_needed(self, table_id: str, partition_desc: str) -> bool:
        """Apply the reference's PG-trigger rule (meta_init.sql:102-150):
        fire when >= 10 versions have accumulated since the last
        CompactionCommit (or since version 0 if never compacted)."""
        cur = self.store.get_latest_partition_info(table_id, partition_desc)
        if cur is None:
            return False
        if cur.commit_op is CommitOp.CompactionCommit:
            return False
        last_compaction = -1
        for p in self.store.get_partition_versions_in_range(
            table_id, partition_desc, 0, cur.version
        ):
            if p.commit_op is CommitOp.CompactionCommit:
                last_compaction = max(last_compaction, p.version)
        if last_compaction >= 0:
            return cur.version - last_compaction >= 10
        return cur.version >= 10

    # ------------------------------------------------------------------ #
    # rollback / cleanup (snapshot management)
    # ------------------------------------------------------------------ #

    def rollback_partition(
        self, table_id: str, partition_desc: str, to_version: int
    ) -> None:
        """Re-commit the snapshot of to_version as the newest version
        (non-destructive rollback, like Spark LakeSoulTable.rollbackPartition)."""
        target = self.store.get_partition_info_by_version(
            table_id, partition_desc, to_version
        )
        if target is None:
            raise ValueError(f"no version {to_version} for {partition_desc}")
        cur = self.store.get_latest_partition_info(table_id, partition_desc)
        nxt = PartitionInfo(
            table_id=table_id,
            partition_desc=partition_desc,
            version=cur.version + 1,
            commit_op=CommitOp.UpdateCommit,
            snapshot=list(target.snapshot),
            domain=target.domain,
        )
        self.store.transaction_insert_partition_info([nxt])
