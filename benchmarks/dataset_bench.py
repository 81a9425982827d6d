#!/usr/bin/env python3
"""PyTorch IterableDataset throughput (BASELINE config 3 at N=1):
merge-on-read scan units decoded into HBM tensors, iterated as training
batches. The 8-GPU DP-shard + RCCL exchange variant is the driver's
SCALE run; this measures the per-GPU feed rate."""
import argparse
import json
import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=100_000_000)
    p.add_argument("--epochs", type=int, default=3)
    p.add_argument("--batch-size", type=int, default=1_000_000)
    args = p.parse_args()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    work = os.environ.get("DS_DIR", "/tmp/dsb")
    os.makedirs(work, exist_ok=True)
    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.io.stream_writer import StreamingWriter
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.entities import CommitOp
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.torch.dataset import LakeSoulIterableDataset

    cat = LakeSoulCatalog(MetaClient(SqliteMetaStore(work + "/meta.db")),
                          warehouse=work + "/wh")
    if cat.table_exists("ds"):
        cat.drop_table("ds", delete_data=True)
    t = cat.create_table(
        "ds", Schema([Field("id", "int64", False), Field("x", "float32"),
                      Field("y", "float32"), Field("lbl", "int32")]),
        hash_bucket_num=32)
    rng = np.random.default_rng(3)
    n = args.rows
    t0 = time.time()
    with StreamingWriter(t, commit_op=CommitOp.AppendCommit,
                         max_rows_per_flush=2_000_000, device=dev) as w:
        done = 0
        while done < n:
            m = min(2_000_000, n - done)
            w.write({"id": np.arange(done, done + m, dtype=np.int64),
                     "x": rng.normal(size=m).astype(np.float32),
                     "y": rng.normal(size=m).astype(np.float32),
                     "lbl": rng.integers(0, 10, m, dtype=np.int32)})
            done += m
    print(f"[ds] wrote {n} rows in {time.time()-t0:.1f}s", file=sys.stderr)

    ds = LakeSoulIterableDataset(t, batch_size=args.batch_size, device=dev)
    # warm epoch (dataset yields dicts of tensors)
    rows = sum(b["id"].numel() for b in ds)
    assert rows == n, rows
    torch.cuda.synchronize() if dev == "cuda" else None
    t0 = time.time()
    for _ in range(args.epochs):
        s = 0.0
        for b in ds:
            # touch the tensors like a training step would
            s += float(b["x"].sum())
    torch.cuda.synchronize() if dev == "cuda" else None
    dt = (time.time() - t0) / args.epochs
    print(json.dumps({"metric": "iterable_dataset_rows_per_sec",
                      "value": n / dt, "rows": n, "s_per_epoch": dt,
                      "batch_size": args.batch_size, "device": dev}))


if __name__ == "__main__":
    main()
