"""Data-asset statistics (reference: lakesoul-flink
``entry/assets/CountDataAssets.java``): rows / bytes / files per table
and per partition, straight from commit metadata (no data scan)."""

from __future__ import annotations

from dataclasses import dataclass
from typing import List


@dataclass
class PartitionStats:
    partition_desc: str
    version: int
    file_count: int
    total_bytes: int


@dataclass
class TableStats:
    table_name: str
    table_id: str
    partitions: List[PartitionStats]

    @property
    def file_count(self) -> int:
        return sum(p.file_count for p in self.partitions)

    @property
    def total_bytes(self) -> int:
        return sum(p.total_bytes for p in self.partitions)


def table_stats(table) -> TableStats:
    parts = []
    for desc in table.partition_descs():
        cur = table.client.store.get_latest_partition_info(table.table_id, desc)
        files = table.client.files_for_partition(table.table_id, desc)
        parts.append(
            PartitionStats(
                partition_desc=desc,
                version=cur.version if cur else -1,
                file_count=len(files),
                total_bytes=sum(f.size for f in files),
            )
        )
    return TableStats(table.info.table_name, table.table_id, parts)
