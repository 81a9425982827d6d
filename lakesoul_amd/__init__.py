"""lakesoul_amd — an MI355X-native Lakehouse IO engine.

A from-scratch reimplementation of the capabilities of lakesoul-io/LakeSoul
(reference: /root/reference) designed for AMD Instinct MI355X (gfx950):

- GPU-resident Parquet scan: host IO + zstd, hand-written HIP kernels for
  PLAIN/dictionary/RLE decode into HBM-resident torch tensors.
- LSM merge-on-read: primary-key sorted delta files merged on GPU
  (radix-sort / merge-path + dedup + segmented merge operators).
- Spark-compatible murmur3-32 (seed 42) hash bucketing, bit-exact with the
  reference (`rust/lakesoul-io/src/utils/hash/`), computed on GPU.
- Scalable metadata with MVCC two-phase commit mirroring the reference's
  PostgreSQL schema (`script/meta_init.sql`); SQLite backend for
  single-node, PostgreSQL backend when a server is available.
- Multi-GPU scan sharding over hash buckets with RCCL (torch.distributed
  "nccl" backend on ROCm) all-to-all shard exchange over xGMI.

The on-disk table format (Parquet files named ``part-{rand}_{bucket:04}.parquet``,
range partition dirs ``col=val/``, ``/compactdir`` compaction convention) and the
metadata commit semantics are kept compatible with the reference.
"""

__version__ = "0.1.0"

from .config import IOConfig  # noqa: F401
from .io.schema import Field, Schema  # noqa: F401
from .meta.client import MetaClient  # noqa: F401
from .tables.catalog import LakeSoulCatalog  # noqa: F401
from .tables.table import LakeSoulTable  # noqa: F401


def execute_sql(catalog, sql, device=None):
    """Run a SQL statement against a catalog (console surface re-export,
    see :mod:`lakesoul_amd.sql`)."""
    from .sql import execute_sql as _run

    return _run(catalog, sql, device=device)
