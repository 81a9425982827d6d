"""Compaction: merge-rewrite a partition's buckets into /compactdir.

Reference behavior: CompactionCommand.scala:330 + CompactBucketIO — read
each bucket through the MOR reader, rewrite as a single compacted file
under ``/compactdir`` (treated as already-merged on read,
merge/mod.rs:358-363), then commit a CompactionCommit whose snapshot
*replaces* the partition's snapshot (metadata_client.rs:585-630). Old
files are recorded in discard_compressed_file_info for deferred cleanup
(CleanExpiredData analog).
"""

from __future__ import annotations

import os
from typing import Optional

from .. import constants
from ..meta.entities import (
    CommitOp,
    DataCommitInfo,
    DataFileOp,
    FileOp,
    MetaInfo,
    PartitionInfo,
)


def compact_partition(table, partition_desc: str, device: Optional[str] = None) -> None:
    from .reader import LakeSoulScan
    from .writer import _write_batch_to_file, random_str

    client = table.client
    cur = client.store.get_latest_partition_info(table.table_id, partition_desc)
    if cur is None:
        return
    scan = LakeSoulScan(table, partitions=[partition_desc], device=device)
    scan.cdc_column = None  # compaction preserves CDC delete markers
    cfg = table.io_config()

    new_ops = []
    old_files = []
    for unit in scan.plan():
        old_files.extend(unit.files)
        if len(unit.files) == 1 and unit.is_compacted_first:
            # already compacted, nothing on top — keep as-is
            new_ops.append(DataFileOp(unit.files[0], FileOp.add))
            continue
        batch = scan._read_unit(unit)
        if batch is None:
            continue
        subdir = (
            "/".join(partition_desc.split(","))
            if partition_desc != constants.NON_PARTITION_TABLE_PART_DESC
            else ""
        )
        out_dir = os.path.join(table.table_path, subdir, constants.COMPACT_DIR)
        os.makedirs(out_dir, exist_ok=True)
        fpath = os.path.join(
            out_dir, f"part-{random_str(16)}_{max(unit.bucket_id, 0):04d}.parquet"
        )
        size = _write_batch_to_file(
            fpath, batch, cfg.compression, cfg.compression_level, cfg.max_row_group_size
        )
        new_ops.append(
            DataFileOp(fpath, FileOp.add, size, ",".join(batch.schema.names()))
        )

    if not new_ops:
        return
    dci = DataCommitInfo(
        table_id=table.table_id,
        partition_desc=partition_desc,
        file_ops=new_ops,
        commit_op=CommitOp.CompactionCommit,
    )
    client.store.insert_data_commit_info(dci)
    client.commit_data(
        MetaInfo(
            table_info=table.info,
            list_partition=[
                PartitionInfo(
                    table_id=table.table_id,
                    partition_desc=partition_desc,
                    snapshot=[dci.commit_id],
                    commit_op=CommitOp.CompactionCommit,
                )
            ],
            read_partition_info=[cur],
        ),
        CommitOp.CompactionCommit,
    )
    # record replaced files for deferred cleanup
    new_paths = {op.path for op in new_ops}
    for fpath in old_files:
        if fpath not in new_paths:
            client.store.insert_discard_file(fpath, table.table_path, partition_desc)


def cleanup_discarded_files(table) -> int:
    """Physically delete files left behind by compaction (reference:
    clean/CleanExpiredData.scala analog). Returns count removed."""
    removed = 0
    for fpath in table.client.store.list_discard_files(table.table_path):
        try:
            if os.path.exists(fpath):
                os.remove(fpath)
            table.client.store.delete_discard_file(fpath)
            removed += 1
        except OSError:
            pass
    return removed
