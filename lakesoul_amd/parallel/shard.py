"""Hash-shard exchange over RCCL (xGMI) — the MI355X-native analog of the
reference's "each engine task reads its own buckets" distribution
(SURVEY.md §2.5): decoded batches are redistributed across the node's
GPUs with all-to-all collectives. xGMI is point-to-point (7 links/GPU),
so pairwise all-to-all saturates aggregate bandwidth where a ring would
bottleneck on one link.

Works with gloo on CPU for tests (world_size>1, one node).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch

from ..io.batch import Batch, Column
from .dist import get_rank_world


def shard_scan(scan):
    """DP-shard a LakeSoulScan by the ambient rank/world (reference:
    arrow/dataset.py:353-394 auto-detection)."""
    rank, world = get_rank_world()
    if world > 1:
        scan.shard(rank, world)
    return scan


def exchange_batch_all_to_all(batch: Batch, dest: torch.Tensor, group=None) -> Batch:
    """Redistribute rows: row i goes to rank ``dest[i]``. Returns the rows
    received by this rank (from all peers, peer-major order).

    Fixed-width columns exchange as one all_to_all_single each; string
    columns exchange lengths then bytes.
    """
    import torch.distributed as dist

    rank, world = get_rank_world()
    if world == 1:
        return batch
    n = batch.num_rows
    dest = dest.to(torch.int64)

    # sort rows by destination so sends are contiguous
    order = torch.argsort(dest, stable=True)
    dest_sorted = dest[order]
    send_counts = torch.bincount(dest_sorted, minlength=world)
    recv_counts = torch.empty_like(send_counts)
    dist.all_to_all_single(recv_counts, send_counts, group=group)
    send_sizes = [int(x) for x in send_counts]
    recv_sizes = [int(x) for x in recv_counts]
    n_recv = sum(recv_sizes)

    out_cols: Dict[str, Column] = {}
    reordered = batch.take(order)
    for f in batch.schema:
        c = reordered.columns[f.name]
        if not c.is_string:
            recv = torch.empty(n_recv, dtype=c.data.dtype, device=c.data.device)
            dist.all_to_all_single(
                recv, c.data.contiguous(), recv_sizes, send_sizes, group=group
            )
            v = None
            if c.validity is not None:
                v = torch.empty(n_recv, dtype=torch.uint8, device=c.validity.device)
                dist.all_to_all_single(v, c.validity.contiguous(), recv_sizes, send_sizes, group=group)
            out_cols[f.name] = Column(f.dtype, data=recv, validity=v)
        else:
            lens = (c.offsets[1:] - c.offsets[:-1]).to(torch.int64)
            recv_lens = torch.empty(n_recv, dtype=torch.int64, device=lens.device)
            dist.all_to_all_single(recv_lens, lens.contiguous(), recv_sizes, send_sizes, group=group)
            # byte splits: sum of lens per destination segment
            byte_send = []
            off = 0
            for s in send_sizes:
                byte_send.append(int(lens[off : off + s].sum()))
                off += s
            byte_recv = []
            off = 0
            for s in recv_sizes:
                byte_recv.append(int(recv_lens[off : off + s].sum()))
                off += s
            recv_bytes = torch.empty(sum(byte_recv), dtype=torch.uint8, device=c.bytes_.device)
            dist.all_to_all_single(
                recv_bytes, c.bytes_.contiguous(), byte_recv, byte_send, group=group
            )
            new_offs = torch.zeros(n_recv + 1, dtype=torch.int64, device=lens.device)
            torch.cumsum(recv_lens, 0, out=new_offs[1:].view(-1))
            v = None
            if c.validity is not None:
                v = torch.empty(n_recv, dtype=torch.uint8, device=c.validity.device)
                dist.all_to_all_single(v, c.validity.contiguous(), recv_sizes, send_sizes, group=group)
            out_cols[f.name] = Column(f.dtype, offsets=new_offs, bytes_=recv_bytes, validity=v)
    return Batch(batch.schema, out_cols)


def rebalance_by_pk(batch: Batch, pk: str, group=None) -> Batch:
    """Re-shard a batch across ranks by spark-murmur3 of a PK column —
    the all-to-all hash-shard exchange used in the multi-GPU dataset path."""
    rank, world = get_rank_world()
    if world == 1:
        return batch
    c = batch.columns[pk]
    if c.data is not None and c.data.device.type == "cuda":
        from ..ops import hip

        empty_prev = torch.empty(0, dtype=torch.int64, device=c.data.device)
        empty_v = torch.empty(0, dtype=torch.uint8, device=c.data.device)
        hashes = hip().hash_fixed_column(c.data, empty_v, empty_prev, True)
        dest = hip().bucket_ids(hashes, world).to(torch.int64)
    else:
        from ..utils.murmur3_np import bucket_ids_np, hash_column

        h = hash_column(c.data.cpu().numpy(), 42)
        dest = torch.from_numpy(bucket_ids_np(h, world).astype("int64"))
    return exchange_batch_all_to_all(batch, dest, group=group)
