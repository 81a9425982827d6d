#!/usr/bin/env python3
"""Phase breakdown of TPC-H q1 through execute_sql (LAKESOUL_TIMING)."""
import os
import sys
import time

os.environ.setdefault("LAKESOUL_TIMING", "1")
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from benchmarks.tpch import Q1_SQL, make_lineitem
from lakesoul_amd.meta.client import MetaClient
from lakesoul_amd.meta.store import SqliteMetaStore
from lakesoul_amd.sql import execute_sql
from lakesoul_amd.tables.catalog import LakeSoulCatalog
from lakesoul_amd.utils import timing


def main():
    device = "cuda" if torch.cuda.is_available() else "cpu"
    work = os.environ.get("TPCH_DIR", "/tmp/q1b")
    os.makedirs(work, exist_ok=True)
    cat = LakeSoulCatalog(MetaClient(SqliteMetaStore(work + "/meta.db")),
                          warehouse=work + "/wh")
    make_lineitem(cat, float(os.environ.get("SF", "1")), device)
    # warm
    execute_sql(cat, Q1_SQL, device=device)
    timing._acc.clear(); timing._cnt.clear()
    steps = int(os.environ.get("STEPS", "5"))
    torch.cuda.synchronize() if device == "cuda" else None
    t0 = time.perf_counter()
    for _ in range(steps):
        execute_sql(cat, Q1_SQL, device=device)
    torch.cuda.synchronize() if device == "cuda" else None
    dt = (time.perf_counter() - t0) / steps
    print(f"q1_sql total: {dt*1000:.1f} ms/query over {steps} steps")
    print(timing.report())


if __name__ == "__main__":
    main()
