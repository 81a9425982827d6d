"""Metadata entities.

Mirrors the reference protobuf entities
(``rust/lakesoul-metadata-proto/src/entity.proto``): TableInfo,
PartitionInfo, DataCommitInfo, DataFileOp, Namespace, and the CommitOp /
FileOp enums with identical numbering (entity.proto:80-99).
"""

from __future__ import annotations

import enum
import json
import time
import uuid as _uuid
from dataclasses import dataclass, field
from typing import List, Optional


class CommitOp(enum.Enum):
    # numbering matches entity.proto:80-91
    CompactionCommit = 0
    AppendCommit = 1
    MergeCommit = 2
    UpdateCommit = 3
    DeleteCommit = 4

    @classmethod
    def from_name(cls, name: str) -> "CommitOp":
        return cls[name]


class FileOp(enum.Enum):
    # entity.proto:94-99
    add = 0
    del_ = 1

    @property
    def text(self) -> str:
        return "add" if self is FileOp.add else "del"

    @classmethod
    def from_text(cls, t: str) -> "FileOp":
        return cls.add if t == "add" else cls.del_


@dataclass
class DataFileOp:
    """entity.proto:102-110; stored in PG as composite type data_file_op."""

    path: str
    file_op: FileOp = FileOp.add
    size: int = 0
    file_exist_cols: str = ""

    def to_json(self) -> dict:
        return {
            "path": self.path,
            "file_op": self.file_op.text,
            "size": self.size,
            "file_exist_cols": self.file_exist_cols,
        }

    @classmethod
    def from_json(cls, d: dict) -> "DataFileOp":
        return cls(
            path=d["path"],
            file_op=FileOp.from_text(d["file_op"]),
            size=int(d.get("size", 0)),
            file_exist_cols=d.get("file_exist_cols", ""),
        )


@dataclass
class Namespace:
    namespace: str
    properties: str = "{}"
    comment: str = ""
    domain: str = "public"


@dataclass
class TableInfo:
    table_id: str
    table_namespace: str = "default"
    table_name: str = ""
    table_path: str = ""
    table_schema: str = ""  # schema JSON (arrow-compatible)
    properties: str = "{}"
    partitions: str = ""  # "rangeCol1,rangeCol2;hashCol1,hashCol2"
    domain: str = "public"

    @staticmethod
    def new_table_id() -> str:
        # reference uses "table_" + uuid (DBUtil)
        return "table_" + str(_uuid.uuid4())

    def get_properties(self) -> dict:
        return json.loads(self.properties) if self.properties else {}

    def hash_bucket_num(self) -> int:
        props = self.get_properties()
        return max(1, int(props.get("hashBucketNum", "1")))

    def primary_keys(self) -> List[str]:
        # partitions format: "rangeKeys;hashKeys" where hashKeys are comma-split
        if ";" not in self.partitions:
            return []
        hash_part = self.partitions.split(";", 1)[1]
        return [c for c in hash_part.split(",") if c]

    def range_keys(self) -> List[str]:
        range_part = self.partitions.split(";", 1)[0]
        return [c for c in range_part.split(",") if c]


@dataclass
class PartitionInfo:
    """partition_info row (meta_init.sql:87-99, entity.proto)."""

    table_id: str
    partition_desc: str
    version: int = -1
    commit_op: CommitOp = CommitOp.AppendCommit
    timestamp: int = 0
    snapshot: List[str] = field(default_factory=list)  # commit UUIDs, in order
    expression: str = ""
    domain: str = "public"


@dataclass
class DataCommitInfo:
    """data_commit_info row (meta_init.sql:71-83)."""

    table_id: str
    partition_desc: str
    commit_id: str = field(default_factory=lambda: str(_uuid.uuid4()))
    file_ops: List[DataFileOp] = field(default_factory=list)
    commit_op: CommitOp = CommitOp.AppendCommit
    committed: bool = False
    timestamp: int = field(default_factory=lambda: int(time.time() * 1000))
    domain: str = "public"


@dataclass
class MetaInfo:
    """entity.proto MetaInfo: the unit handed to commit_data."""

    table_info: TableInfo
    list_partition: List[PartitionInfo] = field(default_factory=list)
    read_partition_info: List[PartitionInfo] = field(default_factory=list)
