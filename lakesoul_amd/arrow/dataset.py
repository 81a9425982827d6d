"""PyArrow-facing dataset view of a LakeSoul table (reference:
``python/src/lakesoul/arrow/dataset.py:43``)."""

from __future__ import annotations

from typing import Iterator, Optional, Sequence


class LakeSoulArrowDataset:
    """Arrow-batch view over a merge-on-read scan, DP-shardable."""

    def __init__(
        self,
        table,
        columns: Optional[Sequence[str]] = None,
        partitions: Optional[Sequence[str]] = None,
        filters: Optional[list] = None,
        batch_size: Optional[int] = None,
        rank: Optional[int] = None,
        world_size: Optional[int] = None,
        device: Optional[str] = None,
    ):
        self.table = table
        self.columns = columns
        self.partitions = partitions
        self.filters = filters
        self.batch_size = batch_size
        self.device = device
        if rank is None or world_size is None:
            from ..parallel.dist import get_rank_world

            rank, world_size = get_rank_world()
        self.rank, self.world_size = rank, world_size

    @property
    def schema(self):
        from ..io.schema import schema_to_arrow

        sel = self.columns or [f.name for f in self.table.schema]
        return schema_to_arrow(self.table.schema.select(sel))

    def _scan(self):
        scan = self.table.scan(
            columns=self.columns,
            partitions=self.partitions,
            filters=self.filters,
            device=self.device,
            batch_size=self.batch_size,
        )
        if self.world_size > 1:
            scan.shard(self.rank, self.world_size)
        return scan

    def to_batches(self) -> Iterator["object"]:
        import pyarrow as pa

        for batch in self._scan().iter_batches():
            for rb in batch.to_arrow().to_batches():
                yield rb

    def to_table(self):
        return self._scan().to_arrow()

    def to_pandas(self):
        return self.to_table().to_pandas()

    def count_rows(self) -> int:
        return self._scan().count()

    def head(self, n: int = 5):
        import pyarrow as pa

        out = []
        got = 0
        for rb in self.to_batches():
            out.append(rb)
            got += rb.num_rows
            if got >= n:
                break
        if not out:
            return self.schema.empty_table()
        return pa.Table.from_batches(out)[:n]
