#!/usr/bin/env python3
"""lakesoul_amd quickstart — the end-to-end user story on one page.

Run anywhere (CPU works; an MI355X accelerates scans automatically):

    python examples/quickstart.py
"""

import os
import sys
import tempfile

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

workdir = tempfile.mkdtemp(prefix="lakesoul_quickstart_")
os.environ["LAKESOUL_META_DB"] = os.path.join(workdir, "meta.db")
os.environ["LAKESOUL_WAREHOUSE"] = os.path.join(workdir, "warehouse")

from lakesoul_amd.io.schema import Field, Schema        # noqa: E402
from lakesoul_amd.sql import execute_sql                # noqa: E402
from lakesoul_amd.tables.catalog import LakeSoulCatalog  # noqa: E402

catalog = LakeSoulCatalog()

# 1. a hash-bucketed primary-key table (Spark-murmur3 bucket layout)
orders = catalog.create_table(
    "orders",
    Schema([
        Field("order_id", "int64", False),
        Field("customer", "string"),
        Field("amount", "decimal(12,2)"),
        Field("qty", "int64"),
    ]),
    primary_keys=["order_id"],
    hash_bucket_num=4,
)

# 2. upserts become sorted delta files; reads merge-on-read (UseLast)
n = 100_000
rng = np.random.default_rng(0)
orders.upsert({
    "order_id": np.arange(n, dtype=np.int64),
    "customer": [f"cust_{i % 1000:04d}" for i in range(n)],
    "amount": rng.integers(100, 10_000_00, n),     # unscaled cents
    "qty": rng.integers(1, 10, n),
})
orders.upsert({                                    # overwrite 10k rows
    "order_id": np.arange(0, n, 10, dtype=np.int64),
    "customer": ["vip"] * (n // 10),
    "amount": np.full(n // 10, 999_99, dtype=np.int64),
    "qty": np.full(n // 10, 1, dtype=np.int64),
})

# 3. scans: filter pushdown (stats + bucket pruning), projections
df = orders.to_pandas(filters=[("order_id", "==", 40)])
print("point lookup:", df.to_dict("records"))

# 4. SQL console surface (joins, aggregates, DML)
print(execute_sql(catalog,
      "SELECT customer, count(*) n, sum(qty) q FROM orders "
      "WHERE customer = 'vip' GROUP BY customer"))
execute_sql(catalog, "UPDATE orders SET qty = 2 WHERE order_id = 40")
print(execute_sql(catalog, "SELECT qty FROM orders WHERE order_id = 40"))

# 5. time travel + compaction + vacuum
print("versions:", orders.latest_version("-5"))
print("v0 row 0:", orders.to_pandas(version=0,
      filters=[("order_id", "==", 0)])["customer"].iloc[0])
orders.compaction()
removed = orders.cleanup_old_versions(keep_latest=1)
print("compacted; vacuumed", removed, "files")

# 6. incremental readers (streaming sources poll new commits)
from lakesoul_amd.tables.stream import TableStream  # noqa: E402

stream = TableStream(orders, device="cpu")
stream.poll()  # drain history (a fresh stream starts from the beginning)
orders.upsert({"order_id": np.array([n + 1], dtype=np.int64),
               "customer": ["new"], "amount": np.array([100], dtype=np.int64),
               "qty": np.array([1], dtype=np.int64)})
batches, advanced = stream.poll()
print("stream delivered", sum(b.num_rows for b in batches), "new rows",
      "(advanced)" if advanced else "")

print("\nquickstart OK —", workdir)
