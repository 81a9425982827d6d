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
        options: Optional[Dict[str, str]] = None,
        incremental: Optional[tuple] = None,
        vector_query: Optional[dict] = None,
    ):
        """Build a LakeSoulScan (reference: catalog.py:740 LakeSoulScan).

        ``vector_query={"column": ..., "query": vec, "k": 10}`` runs an
        ANN search on the column's vector index and restricts the scan to
        the matching PK ids (reference reader.rs:250-331)."""
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
            options=options,
            incremental=incremental,
            vector_query=vector_query,
        )

    def to_arrow(self, **kwargs):
        return self.scan(**kwargs).to_arrow()

    def to_pandas(self, **kwargs):
        return self.to_arrow(**kwargs).to_pandas()

    # -- DML: update / delete (reference: Spark LakeSoulTable.update /
    #    delete, commands/UpdateCommand + DeleteCommand) ----------------- #

    def update(self, filters, assignments: Dict[str, object],
               device: Optional[str] = None) -> int:
        """Update rows matching ``filters``: set ``assignments`` column
        values. PK tables rewrite via upsert (MergeCommit). Returns the
        number of rows updated."""
        if not self.primary_keys:
            raise ValueError("update requires a primary-key table")
        from ..io.batch import concat_batches

        scan = self.scan(filters=filters, device=device)
        batches = list(scan.iter_batches())
        if not batches:
            return 0
        batch = concat_batches(batches)
        import numpy as np
        import torch

        from ..io.batch import Batch, Column, torch_dtype_for

        n = batch.num_rows
        for col, val in assignments.items():
            f = self.schema.field(col)
            if col in self.primary_keys:
                raise ValueError("cannot update a primary-key column")
            if f.is_fixed_width:
                if f.dtype.startswith("decimal") and not isinstance(val, int):
                    # logical literal -> unscaled int64 backing
                    import decimal as _dec

                    from ..io.schema import decimal_params

                    _, sc = decimal_params(f.dtype)
                    val = int(_dec.Decimal(str(val)).scaleb(sc).to_integral_value())
                dev = batch.columns[col].data.device if batch.columns[col].data is not None else "cpu"
                batch.columns[col] = Column(
                    f.dtype,
                    data=torch.full((n,), val, dtype=torch_dtype_for(f.dtype), device=dev),
                )
            else:
                enc = (val.encode() if isinstance(val, str) else bytes(val))
                offs = np.arange(n + 1, dtype=np.int32) * len(enc)
                bys = np.frombuffer(enc * n, dtype=np.uint8).copy() if n else np.empty(0, np.uint8)
                batch.columns[col] = Column(
                    f.dtype, offsets=torch.from_numpy(offs), bytes_=torch.from_numpy(bys)
                )
        self.upsert(batch, device=device)
        return n

    def delete(self, filters=None, device: Optional[str] = None) -> int:
        """Delete matching rows.

        - CDC tables: write delete-marker rows (MergeCommit) — the scan
          drops them (reference CDC semantics).
        - non-CDC PK tables: rewrite each touched bucket without the
          matching rows and commit an UpdateCommit whose snapshot
          replaces the bucket's files.
        - no filters: delete the whole partition (DeleteCommit).
        """
        if filters is None:
            for desc in self.partition_descs():
                self.delete_partition(desc)
            return -1
        props = self.info.get_properties()
        cdc_col = props.get("lakesoul_cdc_change_column")
        from ..io.batch import concat_batches

        if cdc_col:
            scan = self.scan(filters=filters, device=device)
            batches = list(scan.iter_batches())
            if not batches:
                return 0
            batch = concat_batches(batches)
            import numpy as np
            import torch

            from ..io.batch import Column

            n = batch.num_rows
            enc = b"delete"
            offs = np.arange(n + 1, dtype=np.int32) * len(enc)
            bys = np.frombuffer(enc * n, dtype=np.uint8).copy() if n else np.empty(0, np.uint8)
            batch.columns[cdc_col] = Column(
                "string", offsets=torch.from_numpy(offs), bytes_=torch.from_numpy(bys)
            )
            self.upsert(batch, device=device)
            return n
        # rewrite path
        if not self.primary_keys:
            raise ValueError("delete with filters requires a PK or CDC table")
        from ..io.reader import LakeSoulScan
        from ..io.writer import _write_batch_to_file, random_str
        from ..io.filters import resolve_filters
        import os as _os
        import torch

        expr = resolve_filters(filters, self.schema)
        deleted = 0
        cfg = self.io_config()
        full = LakeSoulScan(self, device=device)  # no filters: full rows
        for unit in full.plan():
            batch = full._read_unit(unit)
            if batch is None:
                continue
            keep = ~expr.evaluate(batch)
            n_del = int((~keep).sum())
            if n_del == 0:
                continue
            deleted += n_del
            kept = batch.take(torch.nonzero(keep, as_tuple=True)[0])
            subdir = (
                "/".join(unit.partition_desc.split(","))
                if unit.partition_desc != constants.NON_PARTITION_TABLE_PART_DESC
                else ""
            )
            out_dir = _os.path.join(self.table_path, subdir, constants.COMPACT_DIR)
            _os.makedirs(out_dir, exist_ok=True)
            fpath = _os.path.join(out_dir, f"part-{random_str(16)}_{max(unit.bucket_id, 0):04d}.parquet")
            size = _write_batch_to_file(
                fpath, kept, cfg.compression, cfg.compression_level, cfg.max_row_group_size
            )
            cur = self.client.store.get_latest_partition_info(self.table_id, unit.partition_desc)
            dci = DataCommitInfo(
                table_id=self.table_id,
                partition_desc=unit.partition_desc,
                file_ops=[DataFileOp(fpath, FileOp.add, size, ",".join(kept.schema.names()))]
                + [DataFileOp(p, FileOp.del_) for p in unit.files],
                commit_op=CommitOp.UpdateCommit,
            )
            self.client.store.insert_data_commit_info(dci)
            # snapshot-replace for this bucket: other buckets' files must
            # stay, so resolve current files minus this bucket's + new
            keep_commits = [dci.commit_id]
            cur_files = self.client.files_for_partition(self.table_id, unit.partition_desc)
            other = [f for f in cur_files if f.path not in set(unit.files)]
            if other:
                keep_dci = DataCommitInfo(
                    table_id=self.table_id,
                    partition_desc=unit.partition_desc,
                    file_ops=other,
                    commit_op=CommitOp.UpdateCommit,
                )
                self.client.store.insert_data_commit_info(keep_dci)
                keep_commits = [keep_dci.commit_id, dci.commit_id]
            self.client.commit_data(
                MetaInfo(
                    table_info=self.info,
                    list_partition=[
                        PartitionInfo(
                            table_id=self.table_id,
                            partition_desc=unit.partition_desc,
                            snapshot=keep_commits,
                            commit_op=CommitOp.UpdateCommit,
                        )
                    ],
                    read_partition_info=[cur] if cur else [],
                ),
                CommitOp.UpdateCommit,
            )
        return deleted

    # -- schema evolution (reference: alterTableCommands add columns) --- #

    def add_columns(self, fields) -> None:
        """Append nullable columns to the table schema; existing files
        simply lack them (reads null-fill via exist_cols semantics)."""
        from ..io.schema import Field, Schema, schema_to_json

        schema = self.schema
        for f in fields:
            if not isinstance(f, Field):
                f = Field(*f)
            if not f.nullable:
                raise ValueError("added columns must be nullable")
            if f.name in schema.names():
                raise ValueError(f"column {f.name} already exists")
            schema.fields.append(f)
        self.info.table_schema = schema_to_json(schema)
        self.client.store.update_table_schema(self.table_id, self.info.table_schema)

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

    def cleanup_old_versions(self, keep_latest: int = 1, delete_files: bool = True) -> int:
        """Vacuum: drop partition versions older than the newest
        ``keep_latest`` and delete data files no longer referenced by any
        kept snapshot (reference: CleanExpiredData.scala /
        LakeSoulTable.cleanUpPartitionData). Returns files removed."""
        import os as _os

        removed = 0
        store = self.client.store
        for desc in self.partition_descs():
            cur = store.get_latest_partition_info(self.table_id, desc)
            if cur is None:
                continue
            cutoff = cur.version - keep_latest + 1
            if cutoff <= 0:
                continue
            old = store.get_partition_versions_in_range(self.table_id, desc, 0, cutoff - 1)
            kept = store.get_partition_versions_in_range(
                self.table_id, desc, cutoff, cur.version
            )
            kept_files = set()
            kept_commits = set()
            for p in kept:
                kept_commits.update(p.snapshot)
                for f in self.client._resolve_snapshot_files(self.table_id, desc, p.snapshot):
                    kept_files.add(f.path)
            old_files = set()
            stale_cids = []
            for p in old:
                for f in self.client._resolve_snapshot_files(self.table_id, desc, p.snapshot):
                    old_files.add(f.path)
                for cid in p.snapshot:
                    if cid not in kept_commits:
                        stale_cids.append(cid)
            # one atomic transaction: versions < cutoff + stale commits go
            # together; kept versions are never touched
            store.vacuum_partition_versions(self.table_id, desc, cutoff, stale_cids)
            if delete_files:
                for path in old_files - kept_files:
                    try:
                        if _os.path.exists(path):
                            _os.remove(path)
                        removed += 1
                    except OSError:
                        pass
        return removed

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
