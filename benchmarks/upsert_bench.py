#!/usr/bin/env python3
"""Upsert write-path microbench with phase breakdown (VERDICT r1 #8
scoping: which part of the write would a GPU page-encode kernel
actually move — hash/sort are already GPU, encode+IO are host).

  python benchmarks/upsert_bench.py --rows 20000000 [--device cuda]
"""

import argparse
import json
import os
import sys
import time

import numpy as np
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, REPO)
os.environ.setdefault("LAKESOUL_TIMING", "1")


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--rows", type=int, default=20_000_000)
    p.add_argument("--buckets", type=int, default=16)
    p.add_argument("--reps", type=int, default=3)
    p.add_argument("--device", default="cuda" if torch.cuda.is_available() else "cpu")
    args = p.parse_args()

    from lakesoul_amd.io.schema import Field, Schema
    from lakesoul_amd.meta.client import MetaClient
    from lakesoul_amd.meta.store import SqliteMetaStore
    from lakesoul_amd.tables.catalog import LakeSoulCatalog
    from lakesoul_amd.utils import timing

    wd = os.path.join(os.environ.get("TMPDIR", "/tmp"), "lakesoul_upsert_bench")
    os.makedirs(wd, exist_ok=True)
    catalog = LakeSoulCatalog(
        MetaClient(SqliteMetaStore(os.path.join(wd, "meta.db"))),
        warehouse=os.path.join(wd, "wh"))
    if catalog.table_exists("ub"):
        catalog.drop_table("ub", delete_data=True)
    schema = Schema([
        Field("id", "int64", False), Field("v", "float64", False),
        Field("k", "int32", False), Field("f", "float32", False),
        Field("t", "int64", False),
    ])
    t = catalog.create_table("ub", schema, primary_keys=["id"],
                             hash_bucket_num=args.buckets)
    n = args.rows
    rng = np.random.default_rng(0)
    data = {
        "id": np.arange(n, dtype=np.int64),
        "v": rng.normal(size=n),
        "k": rng.integers(0, 1000, n, dtype=np.int32),
        "f": rng.normal(size=n).astype(np.float32),
        "t": rng.integers(0, 10**12, n, dtype=np.int64),
    }
    row_bytes = 32
    # warm
    t.upsert({k: v[: n // 10] for k, v in data.items()}, device=args.device)
    timing.reset()
    t0 = time.time()
    for _ in range(args.reps):
        t.upsert(data, device=args.device)
    if args.device == "cuda":
        torch.cuda.synchronize()
    dt = (time.time() - t0) / args.reps
    print(json.dumps({
        "rows": n, "reps": args.reps, "device": args.device,
        "s_per_upsert": dt,
        "logical_mb_per_s": n * row_bytes / 1e6 / dt,
    }), flush=True)
    print("[phases]\n" + timing.report(), file=sys.stderr, flush=True)


if __name__ == "__main__":
    main()
