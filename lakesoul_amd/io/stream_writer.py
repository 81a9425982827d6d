"""Streaming writer: feed batches, flush committed files on demand.

Analog of the reference's ``SyncSendableMutableLakeSoulWriter``
(``rust/lakesoul-io/src/writer/mod.rs:250-405``: write_batch /
flush_and_close) and the Flink sink's per-checkpoint file rolling: rows
accumulate per (partition, bucket); when an accumulator exceeds
``max_rows_per_flush`` (or ``max_file_size`` estimate) its file is
written; ``commit()`` writes remaining files and commits all of them in
one DataCommitInfo per partition (two-phase, abort-safe: nothing is
visible until the metadata commit; ``abort()`` deletes written files —
the reference's abort_and_close multipart behavior,
async_writer/mod.rs:80-85)."""

from __future__ import annotations

import os
from typing import Dict, List, Optional

from ..meta.entities import CommitOp, DataCommitInfo, DataFileOp, FileOp, MetaInfo, PartitionInfo
from .batch import Batch, concat_batches
from .writer import FlushResult, write_table_data


class StreamingWriter:
    def __init__(self, table, commit_op: CommitOp = CommitOp.MergeCommit,
                 max_rows_per_flush: int = 2_000_000, device: Optional[str] = None):
        self.table = table
        self.commit_op = commit_op
        self.max_rows = max_rows_per_flush
        self.device = device
        self._pending: List[Batch] = []
        self._pending_rows = 0
        self._results: List[FlushResult] = []
        self._closed = False

    def write(self, data) -> None:
        assert not self._closed
        batch = data if isinstance(data, Batch) else Batch.from_any(data, self.table.schema)
        self._pending.append(batch)
        self._pending_rows += batch.num_rows
        if self._pending_rows >= self.max_rows:
            self.flush()

    def flush(self) -> None:
        if not self._pending:
            return
        batch = concat_batches(self._pending)
        self._pending = []
        self._pending_rows = 0
        self._results.extend(write_table_data(self.table, batch, device=self.device))

    def commit(self) -> List[FlushResult]:
        """Flush remaining rows and atomically commit every written file."""
        self.flush()
        self._closed = True
        if not self._results:
            return []
        by_desc: Dict[str, List[DataFileOp]] = {}
        for r in self._results:
            by_desc.setdefault(r.partition_desc, []).append(
                DataFileOp(r.path, FileOp.add, r.size, r.exist_cols)
            )
        partitions = []
        for desc, ops in by_desc.items():
            dci = DataCommitInfo(
                table_id=self.table.table_id,
                partition_desc=desc,
                file_ops=ops,
                commit_op=self.commit_op,
            )
            self.table.client.store.insert_data_commit_info(dci)
            partitions.append(
                PartitionInfo(
                    table_id=self.table.table_id,
                    partition_desc=desc,
                    snapshot=[dci.commit_id],
                    commit_op=self.commit_op,
                )
            )
        self.table.client.commit_data(
            MetaInfo(table_info=self.table.info, list_partition=partitions),
            self.commit_op,
        )
        return self._results

    def abort(self) -> None:
        """Delete any files written so far; nothing was committed."""
        self._closed = True
        for r in self._results:
            try:
                os.remove(r.path)
            except OSError:
                pass
        self._results = []
        self._pending = []

    def __enter__(self):
        return self

    def __exit__(self, et, ev, tb):
        if et is None:
            self.commit()
        else:
            self.abort()
