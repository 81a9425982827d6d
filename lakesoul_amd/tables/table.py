"""LakeSoulTable — write / upsert / scan / compaction / time-travel.

API surface modeled on the reference's Scala ``LakeSoulTable``
(``lakesoul-spark/.../tables/LakeSoulTable.scala``) and Python
``LakeSoulTable`` (``python/src/lakesoul/catalog.py:303-740``), backed by
the MI355X-native engine.
"""

from __future__ import annotations

import json
from typing import Dict, List, Optional, Sequence, Union

from .. import constants
from ..config import IOConfig
from ..meta.client import MetaClient
from ..meta.entities import (
    CommitOp,
    DataCommitInfo,
    DataFileOp,
    FileOp,
    MetaInfo,
    PartitionInfo,
    TableInfo,
)
from ..io.schema import Schema, schema_from_json


class LakeSoulTable:
    def __init__(self, client: MetaClient, info: TableInfo):
        self.client = client
        self.info = info

    # -- properties ----------------------------------------------------- #

    @property
    def table_id(self) -> str:
        return self.info.table_id

    @property
    def table_path(self) -> str:
        return self.info.table_path

    @property
    def schema(self) -> Schema:
        return schema_from_json(self.info.table_schema)

    @property
    def primary_keys(self) -> List[str]:
        return self.info.primary_keys()

    @property
    def range_keys(self) -> List[str]:
        return self.info.range_keys()

    @property
    def hash_bucket_num(self) -> int:
        return self.info.hash_bucket_num()

    def io_config(self, **overrides) -> IOConfig:
        cfg = IOConfig(
            prefix=self.table_path,
            primary_keys=self.primary_keys,
            range_partitions=self.range_keys,
            hash_bucket_num=self.hash_bucket_num,
        )
        for k, v in overrides.items():
            setattr(cfg, k, v)
        return cfg

    # -- write paths ----------------------------------------------------- #

    def write(self, data, device: Optional[str] = None) -> None:
        """Append (non-PK table) or upsert (PK table) a batch of data.

        ``data``: pyarrow Table/RecordBatch, pandas DataFrame, or a dict of
        numpy arrays / torch tensors.
        """
        if self.primary_keys:
            self.upsert(data, device=device)
        else:
            self._write_commit(data, CommitOp.AppendCommit, device=device)

    def upsert(self, data, device: Optional[str] = None) -> None:
        """Upsert by primary key (reference: LakeSoulTable.scala:273
        executeUpsert -> UpsertCommand with commitType="merge")."""
        if not self.primary_keys:
            raise ValueError("upsert requires a primary-key table")
        self._write_commit(data, CommitOp.MergeCommit, device=device)

    def _write_commit(self, data, commit_op: CommitOp, device: Optional[str] = None) -> None:
        from ..io.writer import write_table_data

        results = write_table_data(self, data, device=device)
        # one DataCommitInfo per partition_desc (reference:
        # TransactionCommit.scala groups new files by partition)
        by_desc: Dict[str, List[DataFileOp]] = {}
        for r in results:
            by_desc.setdefault(r.partition_desc, []).append(
                DataFileOp(
                    path=r.path,
                    file_op=FileOp.add,
                    size=r.size,
                    file_exist_cols=r.exist_cols,
                )
            )
        partitions = []
        for desc, ops in by_desc.items():
            dci = DataCommitInfo(
                table_id=self.table_id,
                partition_desc=desc,
                file_ops=ops,
                commit_op=commit_op,
            )
            self.client.store.insert_data_commit_info(dci)
            partitions.append(
                PartitionInfo(
                    table_id=self.table_id,
                    partition_desc=desc,
                    snapshot=[dci.commit_id],
                    commit_op=commit_op,
                )
            )
        if partitions:
            self.client.commit_data(
                MetaInfo(table_info=self.info, list_partition=partitions), commit_op
            )

    # -- scan paths ------------------------------------------------------ #

    def scan(
        self,
        columns: Optional[Sequence[str]] = None,
        partitions: Optional[Sequence[str]] = None,
        version: Optional[int] = None,
        timestamp_ms: Optional[int] = None,
        filters: Optional[list] = None,
        device: Optional[str] = None,
        batch_size: Optional[int] = None,
    ):
        """Build a LakeSoulScan (reference: catalog.py:740 LakeSoulScan)."""
        from ..io.reader import LakeSoulScan

        return LakeSoulScan(
            self,
            columns=columns,
            partitions=partitions,
            version=version,
            timestamp_ms=timestamp_ms,
            filters=filters,
            device=device,
            batch_size=batch_size,
        )

    def to_arrow(self, **kwargs):
        return self.scan(**kwargs).to_arrow()

    def to_pandas(self, **kwargs):
        return self.to_arrow(**kwargs).to_pandas()

    # -- maintenance ----------------------------------------------------- #

    def compaction(self, partition_desc: Optional[str] = None, device: Optional[str] = None) -> None:
        """Merge-rewrite each bucket of a partition into /compactdir and
        commit a CompactionCommit (reference: CompactionCommand.scala:330,
        compactdir convention merge/mod.rs:358-363)."""
        from ..io.compaction import compact_partition

        descs = (
            [partition_desc]
            if partition_desc is not None
            else self.client.all_partition_descs(self.table_id)
        )
        for desc in descs:
            compact_partition(self, desc, device=device)

    def delete_partition(self, partition_desc: str) -> None:
        cur = self.client.store.get_latest_partition_info(self.table_id, partition_desc)
        if cur is None:
            return
        self.client.commit_data(
            MetaInfo(
                table_info=self.info,
                list_partition=[
                    PartitionInfo(
                        table_id=self.table_id,
                        partition_desc=partition_desc,
                        snapshot=[],
                        commit_op=CommitOp.DeleteCommit,
                    )
                ],
            ),
            CommitOp.DeleteCommit,
        )

    def rollback(self, partition_desc: str, version: int) -> None:
        self.client.rollback_partition(self.table_id, partition_desc, version)

    # -- introspection ---------------------------------------------------- #

    def partition_descs(self) -> List[str]:
        return self.client.all_partition_descs(self.table_id)

    def files(
        self,
        partition_desc: Optional[str] = None,
        version: Optional[int] = None,
        timestamp_ms: Optional[int] = None,
    ) -> List[DataFileOp]:
        descs = (
            [partition_desc]
            if partition_desc is not None
            else self.client.all_partition_descs(self.table_id)
        )
        out: List[DataFileOp] = []
        for d in descs:
            out.extend(
                self.client.files_for_partition(
                    self.table_id, d, version=version, timestamp_ms=timestamp_ms
                )
            )
        return out

    def latest_version(self, partition_desc: str = constants.NON_PARTITION_TABLE_PART_DESC) -> Optional[int]:
        return self.client.get_latest_version(self.table_id, partition_desc)
