"""LakeSoulCatalog — create/drop/list tables and launch scans.

Python API surface modeled on the reference's
``python/src/lakesoul/catalog.py:44-300`` (LakeSoulCatalog) backed by the
MI355X-native engine instead of PyO3/Rust.
"""

from __future__ import annotations

import json
import os
from typing import Dict, List, Optional, Sequence

from ..config import IOConfig
from ..meta.client import MetaClient
from ..meta.entities import TableInfo
from .table import LakeSoulTable


class LakeSoulCatalog:
    def __init__(self, client: Optional[MetaClient] = None, warehouse: Optional[str] = None):
        self.client = client if client is not None else MetaClient()
        self.warehouse = warehouse or os.environ.get(
            "LAKESOUL_WAREHOUSE", os.path.join(os.getcwd(), "lakesoul_warehouse")
        )

    # -- namespaces ----------------------------------------------------- #

    def create_namespace(self, namespace: str) -> None:
        self.client.create_namespace(namespace)

    def list_namespaces(self) -> List[str]:
        return self.client.list_namespaces()

    # -- tables --------------------------------------------------------- #

    def create_table(
        self,
        table_name: str,
        schema: "object",  # pyarrow.Schema or list[(name, dtype_str)]
        primary_keys: Sequence[str] = (),
        range_partitions: Sequence[str] = (),
        hash_bucket_num: int = 1,
        namespace: str = "default",
        table_path: Optional[str] = None,
        properties: Optional[Dict[str, str]] = None,
    ) -> LakeSoulTable:
        from ..io.schema import schema_to_json, normalize_schema

        schema = normalize_schema(schema)
        if table_path is None:
            table_path = os.path.join(self.warehouse, namespace, table_name)
        from ..io.fs import default_fs, is_remote

        if is_remote(table_path):
            default_fs().makedirs(table_path)
        else:
            os.makedirs(table_path, exist_ok=True)
        props = dict(properties or {})
        props.setdefault("hashBucketNum", str(int(hash_bucket_num)))
        partitions = ",".join(range_partitions) + ";" + ",".join(primary_keys)
        info = TableInfo(
            table_id=TableInfo.new_table_id(),
            table_namespace=namespace,
            table_name=table_name,
            table_path=table_path,
            table_schema=schema_to_json(schema),
            properties=json.dumps(props),
            partitions=partitions,
            domain=props.get("domain", "public"),
        )
        self.client.create_table(info)
        return LakeSoulTable(self.client, info)

    def table(self, table_name: str, namespace: str = "default") -> LakeSoulTable:
        info = self.client.get_table_info_by_name(table_name, namespace)
        if info is None:
            raise KeyError(f"table {namespace}.{table_name} not found")
        return LakeSoulTable(self.client, info)

    def table_for_path(self, table_path: str) -> LakeSoulTable:
        info = self.client.get_table_info_by_path(table_path)
        if info is None:
            raise KeyError(f"table at {table_path} not found")
        return LakeSoulTable(self.client, info)

    def table_exists(self, table_name: str, namespace: str = "default") -> bool:
        return self.client.get_table_info_by_name(table_name, namespace) is not None

    def list_tables(self, namespace: str = "default") -> List[str]:
        return [t.table_name for t in self.client.list_tables(namespace)]

    def drop_table(self, table_name: str, namespace: str = "default", delete_data: bool = False) -> None:
        info = self.client.get_table_info_by_name(table_name, namespace)
        if info is None:
            return
        self.client.drop_table(info.table_id)
        if delete_data and info.table_path and os.path.isdir(info.table_path):
            import shutil

            shutil.rmtree(info.table_path, ignore_errors=True)
