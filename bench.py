#!/usr/bin/env python3
"""Flagship benchmark: merge-on-read scan throughput on a hash-PK table.

Reproduces the reference's published scenario (10 M base rows
+ 10 upserts x 2 M rows, batch 8192, zstd(1), dict off) on MI355X:

- setup (untimed): each rank writes its own hash buckets of the synthetic
  table through the GPU write path (murmur3 bucket scatter + PK sort on
  GPU, zstd(1) parquet encode); upsert MB/s is recorded.
- timed step: a full merge-on-read scan of this rank's buckets — host IO
  + zstd decode feeding HIP decode kernels, merge-path k-way merge by PK
  with dedup (UseLast), fused payload gather — plus (for N>1) an RCCL
  all-to-all shard exchange of the merged columns over xGMI.

value = merged output rows/sec aggregated over all N GPUs (whole job).
Weak scaling: per-GPU table shard is fixed as N grows.

Run:  python bench.py --gpus N --steps K --warmup W
(driver launches via torch.distributed.run for N>1; RANK/WORLD_SIZE env)
"""

import argparse
import json
import os
import shutil
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

ROW_BYTES = 8 + 8 + 4 + 4 + 8  # id,v(f64),k(i32),f(f32),t(i64)


def log(rank, msg):
    print(f"[bench rank{rank}] {msg}", flush=True)


def build_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=3)
    p.add_argument("--warmup", type=int, default=1)
    p.add_argument("--rows-base", type=int, default=10_000_000)
    p.add_argument("--rows-upsert", type=int, default=2_000_000)
    p.add_argument("--upserts", type=int, default=10)
    p.add_argument("--buckets-per-gpu", type=int, default=16)
    p.add_argument("--setup-chunk-rows", type=int, default=50_000_000,
                   help="generate+write setup data in chunks to bound host RAM")
    p.add_argument("--device", type=str, default=None)
    p.add_argument("--workdir", type=str, default=None)
    p.add_argument("--keep", action="store_true")
    return p.parse_args()


def main():
    args = build_args()
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    have_gpu = torch.cuda.is_available()
    device = args.device or ("cuda" if have_gpu else "cpu")
    if device == "cuda":
        torch.cuda.set_device(local_rank % max(1, torch.cuda.device_count()))

    dist = None
    if world > 1:
        import torch.distributed as dist_mod

        dist = dist_mod
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29411")
        backend = "nccl" if device == "cuda" else "gloo"
        if device == "cuda":
            # bind the communicator to this rank's GPU (RCCL over xGMI)
            dist.init_process_group(
                backend=backend,
                device_id=torch.device("cuda", local_rank % max(1, torch.cuda.device_count())),
            )
        else:
            dist.init_process_group(backend=backend)

    def barrier():
        if dist is not None:
            dist.barrier()
        if device == "cuda":
            torch.cuda.synchronize()

    workdir = args.workdir or os.path.join(
        os.environ.get("TMPDIR", "/tmp"), f"lakesoul_bench_{world}"
    )
    if rank == 0:
        os.makedirs(workdir, exist_ok=True)
    if dist is not None:
        dist.barrier()

    os.environ["LAKESOUL_META_DB"] = os.path.join(workdir, "meta.db")

    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog

    store = SqliteMetaStore(os.environ["LAKESOUL_META_DB"])
    catalog = LakeSoulCatalog(MetaClient(store), warehouse=os.path.join(workdir, "wh"))

    n_buckets = args.buckets_per_gpu * world
    schema = Schema(
        [
            Field("id", "int64", False),
            Field("v", "float64", False),
            Field("k", "int32", False),
            Field("f", "float32", False),
            Field("t", "int64", False),
        ]
    )

    # ---------------- setup (untimed): write the table ---------------- #
    if rank == 0:
        if catalog.table_exists("bench_mor"):
            catalog.drop_table("bench_mor", delete_data=True)
        catalog.create_table(
            "bench_mor", schema, primary_keys=["id"], hash_bucket_num=n_buckets
        )
    if dist is not None:
        dist.barrier()
    table = catalog.table("bench_mor")

    # every rank generates + writes the rows of ITS buckets (weak scaling:
    # rows per rank fixed)
    rows_base = args.rows_base
    rng = np.random.default_rng(1234 + rank)

    def gen(ids):
        n = len(ids)
        return {
            "id": ids,
            "v": rng.normal(size=n),
            "k": rng.integers(0, 1000, n, dtype=np.int32),
            "f": rng.normal(size=n).astype(np.float32),
            "t": rng.integers(0, 10**12, n, dtype=np.int64),
        }

    my_buckets = set(range(rank * args.buckets_per_gpu, (rank + 1) * args.buckets_per_gpu))

    def filter_my(ids):
        """Keep ids whose murmur3 bucket belongs to this rank."""
        from lakesoul_amd.utils.murmur3_np import bucket_ids_np, hash_column

        h = hash_column(ids, np.uint32(42))
        b = bucket_ids_np(h, n_buckets)
        lo, hi = rank * args.buckets_per_gpu, (rank + 1) * args.buckets_per_gpu
        return ids[(b >= lo) & (b < hi)]

    t0 = time.time()
    upsert_s = 0.0
    upsert_rows = 0
    chunk = max(1, args.setup_chunk_rows)
    for lo in range(0, rows_base * world, chunk):
        hi = min(lo + chunk, rows_base * world)
        base_ids = filter_my(np.arange(lo, hi, dtype=np.int64))
        if len(base_ids) == 0:
            continue
        data = gen(base_ids)
        tu = time.time()
        table.upsert(data, device=device)
        upsert_s += time.time() - tu
        upsert_rows += len(base_ids)
    for u in range(args.upserts):
        up_ids = filter_my(
            rng.choice(rows_base * world, args.rows_upsert * world, replace=False).astype(np.int64)
        )
        data = gen(up_ids)
        tu = time.time()
        table.upsert(data, device=device)
        upsert_s += time.time() - tu
        upsert_rows += len(up_ids)
    setup_s = time.time() - t0
    upsert_mb_s = upsert_rows * ROW_BYTES / 1e6 / upsert_s
    log(rank, f"setup: {upsert_rows} rows written in {setup_s:.1f}s "
              f"(upsert path {upsert_s:.1f}s = {upsert_mb_s:.0f} MB/s logical)")
    if dist is not None:
        dist.barrier()

    # ---------------- timed: MOR scan steps ---------------- #
    def one_scan() -> int:
        """Timed step: MOR scan of this rank's buckets; for N>1 each merged
        batch is re-sharded across ranks by spark-murmur3 of the PK — the
        real rebalance_by_pk exchange (one packed RCCL all-to-all over
        xGMI per batch), enqueued async so it overlaps the next unit's
        decode."""
        from lakesoul_amd.parallel.shard import rebalance_by_pk_async

        scan = table.scan(device=device).shard(rank, world)
        total = 0
        pending = None
        for batch in scan.iter_batches():
            total += batch.num_rows
            if dist is not None:
                if pending is not None:
                    pending.wait()
                pending = rebalance_by_pk_async(batch, "id")
        if pending is not None:
            pending.wait()
        return total

    for w in range(args.warmup):
        one_scan()
    barrier()
    from lakesoul_amd.utils import timing as _tm

    if _tm.ENABLED:
        _tm.reset()  # report warm steps only
    if os.environ.get("LAKESOUL_PROFILE") == "1" and rank == 0:
        import cProfile, pstats, io as _io

        pr = cProfile.Profile()
        pr.enable()
        one_scan()
        pr.disable()
        s = _io.StringIO()
        pstats.Stats(pr, stream=s).sort_stats("cumulative").print_stats(35)
        print(s.getvalue(), file=sys.stderr, flush=True)
    t_start = time.time()
    rows_per_step = 0
    for s in range(args.steps):
        rows_per_step = one_scan()
    barrier()
    elapsed = time.time() - t_start

    # max over ranks
    if dist is not None:
        t_t = torch.tensor([elapsed], dtype=torch.float64)
        if device == "cuda":
            t_t = t_t.cuda()
        dist.all_reduce(t_t, op=dist.ReduceOp.MAX)
        elapsed = float(t_t.item())
        r_t = torch.tensor([float(rows_per_step)], dtype=torch.float64)
        if device == "cuda":
            r_t = r_t.cuda()
        dist.all_reduce(r_t, op=dist.ReduceOp.SUM)
        total_rows_per_step = float(r_t.item())
        u_t = torch.tensor([upsert_mb_s], dtype=torch.float64)
        if device == "cuda":
            u_t = u_t.cuda()
        dist.all_reduce(u_t, op=dist.ReduceOp.SUM)
        total_upsert_mb_s = float(u_t.item())
    else:
        total_rows_per_step = float(rows_per_step)
        total_upsert_mb_s = upsert_mb_s

    ms_per_step = elapsed / args.steps * 1000.0
    value = total_rows_per_step / (elapsed / args.steps)

    from lakesoul_amd.utils import timing as _timing

    if _timing.ENABLED and rank == 0:
        print("[timing]\n" + _timing.report(), file=sys.stderr, flush=True)

    if rank == 0:
        result = {
            "metric": "mor_scan_rows_per_sec",
            "value": value,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "mixed(int64,float64,int32,float32)",
            "data": "synthetic",
            "config": {
                "model": "hash-PK lakehouse table, merge-on-read scan",
                "scenario": "10M base + 10x2M upserts per GPU (reference BASELINE.md config)",
                "rows_base_per_gpu": args.rows_base,
                "upserts": args.upserts,
                "rows_upsert_per_gpu": args.rows_upsert,
                "hash_buckets": n_buckets,
                "batch": "row-group<=250k, zstd(1), dict off",
                "global_batch": rows_per_step,
                "seq_len": None,
                "parallelism": f"dp{world} hash-bucket sharding + RCCL all-to-all exchange",
                "merged_rows_per_step_per_gpu": rows_per_step,
                "upsert_mb_per_s_total": total_upsert_mb_s,
                "device": device,
            },
        }
        print(json.dumps(result), flush=True)

    if dist is not None:
        dist.destroy_process_group()
    if rank == 0 and not args.keep and args.workdir is None:
        shutil.rmtree(workdir, ignore_errors=True)


if __name__ == "__main__":
    main()
