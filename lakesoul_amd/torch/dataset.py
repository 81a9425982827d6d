"""PyTorch IterableDataset over a LakeSoul table.

MI355X-native analog of the reference's ``python/src/lakesoul/torch/
dataset.py:15-20``: merge-on-read scan units (partition x hash-bucket)
are DP-sharded ``i % world_size == rank`` (arrow/dataset.py:353-394
semantics), decoded straight into HBM-resident torch tensors on the GPU
path, and optionally re-sharded across the node's GPUs with an RCCL
all-to-all over xGMI (parallel/shard.py).
"""

from __future__ import annotations

from typing import Dict, Iterator, Optional, Sequence

import torch
from torch.utils.data import IterableDataset

from ..parallel.dist import get_rank_world


class LakeSoulIterableDataset(IterableDataset):
    def __init__(
        self,
        table,
        columns: Optional[Sequence[str]] = None,
        partitions: Optional[Sequence[str]] = None,
        batch_size: Optional[int] = None,
        device: Optional[str] = None,
        filters: Optional[list] = None,
        exchange_by: Optional[str] = None,
        yield_arrow: bool = False,
    ):
        self.table = table
        self.columns = columns
        self.partitions = partitions
        self.batch_size = batch_size
        self.device = device
        self.filters = filters
        self.exchange_by = exchange_by
        self.yield_arrow = yield_arrow

    def _scan(self):
        scan = self.table.scan(
            columns=self.columns,
            partitions=self.partitions,
            filters=self.filters,
            device=self.device,
            batch_size=self.batch_size,
        )
        rank, world = get_rank_world()
        # compose DataLoader worker sharding with DP rank sharding
        info = torch.utils.data.get_worker_info()
        if info is not None and info.num_workers > 1:
            scan.shard(rank * info.num_workers + info.id, world * info.num_workers)
        elif world > 1:
            scan.shard(rank, world)
        return scan

    def __iter__(self) -> Iterator[Dict[str, torch.Tensor]]:
        for batch in self._scan().iter_batches():
            if self.exchange_by is not None:
                from ..parallel.shard import rebalance_by_pk

                batch = rebalance_by_pk(batch, self.exchange_by)
            if self.yield_arrow:
                yield batch.to_arrow()
            else:
                out = {}
                for f in batch.schema:
                    c = batch.columns[f.name]
                    if c.is_string:
                        out[f.name] = (c.offsets, c.bytes_)
                    else:
                        out[f.name] = c.data
                yield out
