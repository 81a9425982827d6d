import sys, faulthandler
faulthandler.enable()
sys.path.insert(0, "/root/repo")
import numpy as np
from lakesoul_amd.meta.client import MetaClient
from lakesoul_amd.meta.store import SqliteMetaStore
from lakesoul_amd.tables.catalog import LakeSoulCatalog
from lakesoul_amd.io.schema import Field, Schema
import tempfile
d = tempfile.mkdtemp()
cat = LakeSoulCatalog(MetaClient(SqliteMetaStore(d+"/m.db")), warehouse=d+"/wh")
t = cat.create_table(
    "grange",
    Schema([Field("dt", "string", False), Field("id", "int64", False),
            Field("v", "float64", False)]),
    primary_keys=["id"], range_partitions=["dt"], hash_bucket_num=2)
n = 40000
for day in ("2026-01-01", "2026-01-02"):
    t.upsert({"dt": [day] * n, "id": np.arange(n, dtype=np.int64), "v": np.zeros(n)})
    t.upsert({"dt": [day] * (n // 4), "id": np.arange(0, n, 4, dtype=np.int64),
              "v": np.ones(n // 4)})
print("WRITTEN", flush=True)
cpu = t.scan(device="cpu", partitions=["dt=2026-01-02"]).to_arrow().to_pandas()
print("CPU_OK", len(cpu), flush=True)
gpu = t.scan(device="cuda", partitions=["dt=2026-01-02"]).to_arrow().to_pandas()
print("GPU_OK", len(gpu), flush=True)
