"""Scan path: file planning + decode + merge-on-read.

MI355X-native analog of the reference's ``LakeSoulReader``
(``rust/lakesoul-io/src/reader.rs``) + ``MergeParquetExec``
(``physical_plan/merge/mod.rs``):

- resolve the file list per (partition, hash bucket) from metadata;
- PK point-filter bucket pruning (reader.rs:164-225);
- pass-through for compacted/single files (merge/mod.rs:291-377);
- sorted merge by PK across delta files with merge operators;
- schema evolution: files missing a requested column yield nulls
  (DefaultColumnStream analog, stream helpers).

On GPU the decode + merge run as HIP kernels over HBM-resident columns
(csrc/hip/kernels.hip); the CPU path (numpy) is the correctness oracle.
"""

from __future__ import annotations

import os
import re
from dataclasses import dataclass, field
from typing import Dict, Iterator, List, Optional, Sequence, Tuple

import numpy as np
import torch

from .. import constants
from ..ops import cpp
from .batch import Batch, Column
from .filters import decode_stat, resolve_filters
from .merge_cpu import NpColumn, merge_sorted_files
from .schema import FIXED_WIDTH_BYTES, Schema

_BUCKET_RE = re.compile(r"part-[^/]*_(\d+)\.[a-zA-Z0-9]+$")

_NP_FROM_PHYS = {
    "bool": np.uint8,
    "int8": np.int32,
    "int16": np.int32,
    "int32": np.int32,
    "int64": np.int64,
    "float32": np.float32,
    "float64": np.float64,
    "date32": np.int32,
    "timestamp[us]": np.int64,
    "timestamp[ms]": np.int64,
    "timestamp[ns]": np.int64,
}

def _np_phys(dtype):
    return np.int64 if dtype.startswith("decimal") else _NP_FROM_PHYS[dtype]


def _np_target(dtype):
    return np.int64 if dtype.startswith("decimal") else _NP_TARGET[dtype]


_NP_TARGET = {
    "bool": np.uint8,
    "int8": np.int8,
    "int16": np.int16,
    "int32": np.int32,
    "int64": np.int64,
    "float32": np.float32,
    "float64": np.float64,
    "date32": np.int32,
    "timestamp[us]": np.int64,
    "timestamp[ms]": np.int64,
    "timestamp[ns]": np.int64,
}


def extract_hash_bucket_id(path: str) -> Optional[int]:
    """Recover bucket id from ``part-{rand}_{bucket:04}.parquet``
    (reference: helpers/mod.rs:926-945)."""
    m = _BUCKET_RE.search(path)
    return int(m.group(1)) if m else None


@dataclass
class ScanUnit:
    partition_desc: str
    bucket_id: int
    files: List[str]  # snapshot order (oldest -> newest)
    is_compacted_first: bool = False


class LakeSoulScan:
    def __init__(
        self,
        table,
        columns: Optional[Sequence[str]] = None,
        partitions: Optional[Sequence[str]] = None,
        version: Optional[int] = None,
        timestamp_ms: Optional[int] = None,
        filters: Optional[list] = None,
        device: Optional[str] = None,
        batch_size: Optional[int] = None,
        incremental: Optional[Tuple[int, int]] = None,
        options: Optional[dict] = None,
        vector_query: Optional[dict] = None,
    ):
        self.table = table
        self.schema: Schema = table.schema
        self.pk = table.primary_keys
        self.range_cols = table.range_keys
        sel = list(columns) if columns else [f.name for f in self.schema]
        self.out_schema = self.schema.select(sel)
        # merge needs PK columns even if not selected; range-partition
        # columns are not stored in files (materialized from partition_desc)
        self.read_cols = [
            n
            for n in dict.fromkeys((self.pk + sel) if self.pk else sel)
            if n not in self.range_cols
        ]
        self.partitions = list(partitions) if partitions else None
        self.version = version
        self.timestamp_ms = timestamp_ms
        self.filter_expr = resolve_filters(filters, self.schema)
        # ANN-result injection: a vector_query turns into a PK id filter
        # inside the normal scan (reference reader.rs:250-331
        # inject_vector_search_filter)
        self.vector_result = None
        if vector_query is not None:
            self._inject_vector_filter(dict(vector_query))
        # filter columns must be read (and materialized) even when not
        # selected; the final projection drops them (session.rs:650-730
        # projection computation analog)
        if self.filter_expr is not None:
            for c in sorted(self.filter_expr.columns()):
                if c not in self.read_cols and c not in self.range_cols:
                    self.read_cols.append(c)
        # struct/map columns read as their parquet leaves (struct s ->
        # s.a / s.b; map m -> m.key / m.value parallel lists — the same
        # dotted names flatten_element surfaces); the MOR merge runs per
        # leaf and _np_to_batch reassembles the top-level column
        from .schema import Field as _LsField, map_params, struct_members

        self._leaf_fields: Dict[str, "object"] = {}
        expanded: List[str] = []
        for n in self.read_cols:
            fld = self.schema.field(n)
            sm = struct_members(fld.dtype)
            mp = map_params(fld.dtype)
            if sm is not None:
                for mn, mt in sm:
                    ln = f"{n}.{mn}"
                    self._leaf_fields[ln] = _LsField(ln, mt, fld.nullable)
                    expanded.append(ln)
            elif mp is not None:
                kt, vt = mp
                for role, t in (("key", f"list<{kt}>"),
                                ("value", f"list<{vt}>")):
                    ln = f"{n}.{role}"
                    self._leaf_fields[ln] = _LsField(ln, t, fld.nullable)
                    expanded.append(ln)
            else:
                expanded.append(n)
        self.read_cols = expanded
        self.eval_fields = list(self.out_schema.fields)
        have = {f.name for f in self.eval_fields}
        if self.filter_expr is not None:
            for c in sorted(self.filter_expr.columns()):
                if c not in have:
                    self.eval_fields.append(self.schema.field(c))
        self.eval_schema = Schema(self.eval_fields)
        self.device = str(device or table.io_config().resolve_device())
        if self.device.startswith("cuda"):
            # normalize "cuda:N" -> select device N, compare as "cuda"
            import torch as _torch

            if ":" in self.device and _torch.cuda.is_available():
                _torch.cuda.set_device(self.device)
            self.device = "cuda"
        opts = dict(options or {})
        cfg_opt = opts.get("scan_cache", table.io_config().option("scan_cache", "0"))
        self.use_cache = str(cfg_opt) == "1"
        self.batch_size = batch_size
        self.incremental = incremental
        self._shard: Optional[Tuple[int, int]] = None
        props = table.info.get_properties()
        self.cdc_column = props.get("lakesoul_cdc_change_column", None)
        # CDC filtering needs the change column even when the projection
        # excludes it (the merge drops 'delete' rows); eval_schema keeps
        # the user's projection, so it never reaches the output
        if self.cdc_column and self.cdc_column not in self.read_cols:
            self.read_cols.append(self.cdc_column)
        self.merge_ops = dict(table.io_config().merge_operators)
        for k, v in props.items():
            if k.startswith("merge_op."):
                self.merge_ops[k[len("merge_op."):]] = v

    # ------------------------------------------------------------------ #

    def _inject_vector_filter(self, vq: dict) -> None:
        """Run the ANN search and AND `pk IN (top-k ids)` into the scan
        filter. vq keys: column (str), query (array, 1-D or (nq, dim)),
        k (int, default 10), nprobe, rescore, device."""
        import numpy as _np

        from ..vector.index import VectorIndex

        column = vq.pop("column")
        query = vq.pop("query")
        k = int(vq.pop("k", 10))
        if len(self.pk) != 1:
            raise ValueError("vector_query needs a single-PK table")
        root = os.path.join(self.table.table_path, "_vector_index", column)
        if not os.path.exists(os.path.join(root, "manifest.json")):
            raise ValueError(
                f"no vector index for column {column!r}; call "
                "build_vector_index first")
        idx = VectorIndex.load(root)
        kwargs = {}
        for opt in ("nprobe", "rescore"):
            if vq.get(opt) is not None:
                kwargs[opt] = vq[opt]
        dev = vq.get("device") or self.table.io_config().resolve_device()
        ids, scores = idx.search(query, k=k, device=str(dev), **kwargs)
        self.vector_result = (ids, scores)
        flat = sorted(set(int(i) for i in _np.asarray(ids).ravel() if i >= 0))
        from .filters import And as _And
        from .filters import Cmp as _Cmp

        id_filter = _Cmp(self.pk[0], "in", flat)
        self.filter_expr = (id_filter if self.filter_expr is None
                            else _And(self.filter_expr, id_filter))

    def shard(self, rank: int, world_size: int) -> "LakeSoulScan":
        """DP sharding: scan unit i -> rank i % world_size (reference:
        python arrow/dataset.py:353-394)."""
        self._shard = (rank, world_size)
        return self

    def plan(self) -> List[ScanUnit]:
        client = self.table.client
        tid = self.table.table_id
        descs = self.partitions or client.all_partition_descs(tid)
        # range-partition pruning from the filter expression (the
        # reference's metadata partition filtering, helpers/mod.rs)
        if self.filter_expr is not None:
            kept = []
            for desc in descs:
                pv = {}
                for kv in desc.split(","):
                    if "=" in kv:
                        k, v = kv.split("=", 1)
                        pv[k] = constants.decode_partition_value(v)
                if self.filter_expr.partition_prune(pv):
                    kept.append(desc)
            descs = kept
        units: List[ScanUnit] = []
        for desc in descs:
            if self.incremental is not None:
                file_ops = client.incremental_files(tid, desc, *self.incremental)
            else:
                file_ops = client.files_for_partition(
                    tid, desc, version=self.version, timestamp_ms=self.timestamp_ms
                )
            by_bucket: Dict[int, List[str]] = {}
            for op in file_ops:
                # invalid-file tolerance (reference session.rs:440-450:
                # listing drops files < 8 bytes with an error log)
                from .fs import is_remote as _is_remote

                if not _is_remote(op.path):
                    try:
                        if os.path.getsize(op.path) < 8:
                            import warnings

                            warnings.warn(f"skipping invalid file {op.path}")
                            continue
                    except OSError:
                        import warnings

                        warnings.warn(f"skipping missing file {op.path}")
                        continue
                b = extract_hash_bucket_id(op.path)
                # bucket_id -1 = unknown (file name lacks the part-*_NNNN
                # suffix, e.g. foreign/imported files): exempt from bucket
                # pruning below
                by_bucket.setdefault(b if b is not None else -1, []).append(op.path)
            for b, files in sorted(by_bucket.items()):
                if self.filter_expr is not None:
                    files = self.prune_files_by_stats(files)
                    if not files:
                        continue
                is_comp = constants.COMPACT_DIR in files[0].split(os.sep) if files else False
                units.append(ScanUnit(desc, b, files, is_comp))
        # PK point-filter bucket pruning (reader.rs:164-225)
        pruned_bucket = self._bucket_filter()
        if pruned_bucket is not None:
            units = [u for u in units
                     if u.bucket_id < 0 or u.bucket_id in pruned_bucket]
        if self._shard is not None:
            rank, ws = self._shard
            units = [u for i, u in enumerate(units) if i % ws == rank]
        return units

    def _bucket_filter(self) -> Optional[set]:
        """If filters pin every PK column to constants, only matching
        buckets need scanning (reference: reader.rs:164-225)."""
        if not self.pk or self.filter_expr is None:
            return None
        eq = {
            k: v
            for k, v in self.filter_expr.pk_eq_values().items()
            if k in self.pk
        }
        if set(eq.keys()) != set(self.pk):
            return None
        from ..utils import murmur3 as m3

        h = 0
        for i, name in enumerate(self.pk):
            dt = self.schema.field(name).dtype
            seed = m3.HASH_SEED if i == 0 else h
            h = m3.hash_value(eq[name], dt, seed)
        return {h % self.table.hash_bucket_num}

    # ------------------------------------------------------------------ #

    def _field_for(self, name: str):
        """Resolve a read column: schema field, or a synthesized leaf
        field for struct members / map key-value lists."""
        lf = self._leaf_fields.get(name)
        return lf if lf is not None else self.schema.field(name)

    def _has_list_str(self) -> bool:
        """list<string> (incl. map<string,*>/map<*,string> leaves that
        expand to it) decodes on the host (prefixed-stream MOR) and
        ships to HBM — it bypasses the GPU unit path. Struct members and
        primitive map leaves are flat/list columns the GPU path handles
        natively."""
        return any(self._field_for(n).dtype == "list<string>"
                   for n in self.read_cols if n not in self.range_cols)

    def iter_batches(self) -> Iterator[Batch]:
        units = self.plan()
        if (self.device == "cuda" and len(units) > 1
                and not self._has_list_str()
                and self._gpu_merge_supported() and not self.use_cache
                and all(self._unit_fits(u) for u in units)
                and os.environ.get("LAKESOUL_SCAN_PIPELINE", "1") != "0"):
            gen = self._iter_units_pipelined(units)
        else:
            gen = (self._read_unit(u) for u in units)
        for batch in gen:
            if batch is None:
                continue
            batch = self._apply_filters(batch)
            if len(self.eval_schema) != len(self.out_schema):
                batch = Batch(
                    self.out_schema,
                    {f.name: batch.columns[f.name] for f in self.out_schema},
                )
            if self.batch_size:
                n = batch.num_rows
                for off in range(0, n, self.batch_size):
                    # contiguous rows: zero-copy slice views, not a gather
                    yield batch.slice(off, min(off + self.batch_size, n))
            else:
                yield batch

    def _iter_units_pipelined(self, units: List[ScanUnit],
                              depth: Optional[int] = None):
        if depth is None:
            # A/B on MI355X (benchmarks/scan_depth_ab.py, 2026-09-12):
            # the step is host-fetch bound and deeper prefetch keeps the
            # 16-thread pool busy across unit boundaries — 3: 47.1,
            # 8: 36.6, 10: 36.4, 16: 37.8 ms/step; 2 HIP streams beat
            # 3-4 (extra streams fragment the copy queue)
            depth = int(os.environ.get("LAKESOUL_SCAN_DEPTH", "10"))
        """GPU path, two-level pipeline (overlap engineering, SURVEY.md
        §7.2 item 5):
        - host stage: prefetch IO/decompress of upcoming units on
          ``depth`` threads (C++ releases the GIL);
        - GPU stage: two worker threads, each with its own HIP stream,
          run H2D + decode + merge for alternating units — unit k+1's
          transfers and kernels overlap unit k's (each unit ends with an
          unavoidable sync when the merged row count materializes).
        Results are yielded in plan order."""
        from concurrent.futures import ThreadPoolExecutor

        import torch as _torch

        from .reader_gpu import fetch_raw, read_unit_gpu

        nstreams = int(os.environ.get("LAKESOUL_SCAN_STREAMS", "2"))
        streams = [_torch.cuda.Stream() for _ in range(max(1, nstreams))]

        def proc(i, unit, fetch_fut):
            raw = fetch_fut.result()  # resolve on the GPU worker thread
            s = streams[i % len(streams)]
            with _torch.cuda.stream(s):
                batch = read_unit_gpu(self, unit, raw)
            s.synchronize()
            # the batch's tensors were ALLOCATED on side stream s but are
            # consumed (and eventually freed) on the default stream: tag
            # the default stream as a user so the caching allocator does
            # not hand their blocks back to s-stream work that may still
            # overlap the consumer's reads (classic record_stream
            # cross-stream lifetime rule; without this, later units
            # reusing s corrupt still-referenced batches)
            cur = _torch.cuda.current_stream()
            if batch is not None:
                for c in batch.columns.values():
                    for t in (c.data, c.offsets, c.bytes_, c.validity):
                        if t is not None and t.is_cuda:
                            t.record_stream(cur)
            return batch

        with ThreadPoolExecutor(max_workers=depth) as fex, ThreadPoolExecutor(
            max_workers=len(streams)
        ) as gex:
            fetches = [
                fex.submit(fetch_raw, self._localize(u.files), self.read_cols)
                for u in units[: depth]
            ]
            procs: dict = {}
            for i in range(min(len(streams), len(units))):
                procs[i] = gex.submit(proc, i, units[i], fetches[i])
            for i, unit in enumerate(units):
                if i + depth < len(units):
                    fetches.append(
                        fex.submit(
                            fetch_raw,
                            self._localize(units[i + depth].files),
                            self.read_cols,
                        )
                    )
                j = i + len(streams)
                if j < len(units):
                    procs[j] = gex.submit(proc, j, units[j], fetches[j])
                yield procs.pop(i).result()

    def __iter__(self):
        return self.iter_batches()

    def to_arrow(self):
        import pyarrow as pa

        from .schema import schema_to_arrow

        tables = [b.to_arrow() for b in self.iter_batches()]
        if not tables:
            return schema_to_arrow(self.out_schema).empty_table()
        return pa.concat_tables(tables)

    def to_batch(self) -> Batch:
        """Whole scan as one Batch (tensor columns; stays on the scan
        device — the query engine's input)."""
        batches = list(self.iter_batches())
        if not batches:
            import numpy as _np

            cols = {}
            for f in self.out_schema:
                if f.dtype in ("string", "binary"):
                    cols[f.name] = Column(
                        f.dtype, offsets=torch.zeros(1, dtype=torch.int64),
                        bytes_=torch.empty(0, dtype=torch.uint8))
                else:
                    from .batch import torch_dtype_for

                    cols[f.name] = Column(
                        f.dtype, data=torch.empty(0, dtype=torch_dtype_for(f.dtype)))
            return Batch(self.out_schema, cols)
        if len(batches) == 1:
            return batches[0]
        from .batch import concat_batches

        return concat_batches(batches)

    def count(self) -> int:
        """Count-only fast path (EmptyScanCountExec analog,
        physical_plan/empty_schema.rs:192): row counts come from parquet
        footers; PK tables still need the merge for dedup."""
        if not self.pk and self.filter_expr is None:
            total = 0
            for unit in self.plan():
                for f in self._localize(unit.files):
                    h = cpp().open_parquet(f)
                    try:
                        total += cpp().parquet_meta(h)["num_rows"]
                    finally:
                        cpp().close_parquet(h)
            return total
        return sum(b.num_rows for b in self.iter_batches())

    # ------------------------------------------------------------------ #

    def _localize(self, files: List[str]) -> List[str]:
        from .fs import default_fs, is_remote

        if not any(is_remote(p) for p in files):
            return files
        fs = default_fs()
        return [fs.localize(p) if is_remote(p) else p for p in files]

    def _gpu_merge_supported(self) -> bool:
        """All PK types merge on the GPU now: integers via packed-u64
        merge-path, strings via stable LSD passes over 8-byte chunk keys
        (merge_gpu._string_sort_tensors). LAKESOUL_GPU_STRING_MERGE=0
        forces the hybrid CPU-merge path for string PKs."""
        if not self.pk:
            return True
        if os.environ.get("LAKESOUL_GPU_STRING_MERGE", "1") == "0":
            return all(
                self.schema.field(p).dtype not in ("string", "binary")
                for p in self.pk
            )
        return True

    def _unit_fits(self, unit: ScanUnit) -> bool:
        """Estimate whether a bucket's decoded size fits the configured
        device-memory budget (LAKESOUL_MAX_UNIT_BYTES, default 64 GB —
        leaves headroom in 288 GB HBM for merge intermediates)."""
        limit = int(os.environ.get("LAKESOUL_MAX_UNIT_BYTES", str(64 * 1024**3)))
        total = 0
        for path in unit.files:
            try:
                total += os.path.getsize(path)
            except OSError:
                pass
        # decompressed estimate: zstd(1) on typical columns ~2x
        return total * 2 <= limit

    def _read_unit(self, unit: ScanUnit) -> Optional[Batch]:
        if not unit.files:
            return None
        oversized = not self._unit_fits(unit)
        if oversized and self.device != "cuda" and not self._chunkable():
            raise MemoryError(
                f"scan unit bucket={unit.bucket_id} estimated decoded size "
                f"exceeds LAKESOUL_MAX_UNIT_BYTES. Recreate the table with "
                f"more hash buckets, compact the partition, or raise the limit."
            )
        cache_key = None
        if self.use_cache:
            from .hbm_cache import scan_cache

            cache_key = scan_cache().key(unit, self.read_cols)
            hit = scan_cache().get(cache_key)
            if hit is not None:
                return hit
        unit = ScanUnit(unit.partition_desc, unit.bucket_id,
                        self._localize(unit.files), unit.is_compacted_first)
        needs_merge = bool(self.pk) and (
            len(unit.files) > 1 or self.cdc_column is not None or bool(self.merge_ops)
        )
        if oversized and self._chunkable():
            try:
                return self._read_unit_chunked(unit)
            except MemoryError:
                raise
            except Exception as e:
                # e.g. foreign files without PK stats: fall through to the
                # non-chunked degraded paths below
                import warnings

                warnings.warn(f"chunked merge unavailable ({e}); "
                              "falling back")
        if self.device == "cuda" and (
            oversized
            or self._has_list_str()
            or (needs_merge and not self._gpu_merge_supported())
        ):
            # hybrid: CPU decode+merge (host RAM), then ship the merged
            # batch to HBM — for string-PK merges and for buckets whose
            # working set would not fit the HBM budget (the merged output
            # is far smaller than decode intermediates)
            if oversized:
                import warnings

                warnings.warn(
                    f"scan unit bucket={unit.bucket_id} exceeds "
                    "LAKESOUL_MAX_UNIT_BYTES; falling back to host-side "
                    "decode+merge for this bucket (slower). Compact or "
                    "re-bucket the table to restore the GPU path."
                )
            batch = self._read_unit_cpu(unit)
            return batch.to_device("cuda") if batch is not None else None
        if self.device == "cuda":
            from .reader_gpu import read_unit_gpu

            batch = read_unit_gpu(self, unit)
        else:
            batch = self._read_unit_cpu(unit)
        if cache_key is not None and batch is not None:
            from .hbm_cache import scan_cache

            scan_cache().put(cache_key, batch)
        return batch

    def _chunkable(self) -> bool:
        """Chunked (PK-range) merge needs a single primary key whose
        row-group min/max stats order the key space: integers, or
        strings (lexicographic — trusts exact, untruncated byte-array
        stats, which our writer always records)."""
        return (len(self.pk) == 1 and
                self.schema.field(self.pk[0]).dtype in
                ("int64", "int32", "int16", "int8", "string"))

    def _read_unit_chunked(self, unit: ScanUnit) -> Optional[Batch]:
        """Chunked spill merge for buckets larger than the memory budget
        (ROADMAP item): split the PK space at row-group stat boundaries,
        decode+merge only the row groups overlapping each range, filter
        the range's rows, concatenate. Works because bucket files are
        globally PK-sorted, so ranges partition PK groups exactly."""
        import warnings

        from .batch import concat_batches

        limit = int(os.environ.get("LAKESOUL_MAX_UNIT_BYTES", str(64 * 1024**3)))
        pk0 = self.pk[0]
        files = self._localize(unit.files)
        # per-(file,rg): pk min/max + estimated decoded bytes
        spans = []   # (path, rg, lo, hi, est_bytes)
        from .batch import np_dtype_for

        row_bytes = sum(
            np.dtype(np_dtype_for(f.dtype)).itemsize if f.is_fixed_width
            else 16   # strings: offsets + typical payload
            for f in self.eval_schema
        ) or 8
        for path in files:
            h = cpp().open_parquet(path)
            try:
                meta = cpp().parquet_meta(h)
                cols = meta["columns"]
                pk_ci = next(i for i, c in enumerate(cols) if c["name"] == pk0)
                dtype = cols[pk_ci]["dtype"]
                for rg, g in enumerate(meta["row_groups"]):
                    sd = g["columns"][pk_ci]
                    if "min" not in sd:
                        raise ValueError("missing pk stats")
                    lo = decode_stat(sd["min"], dtype)
                    hi = decode_stat(sd["max"], dtype)
                    spans.append((path, rg, lo, hi, g["num_rows"] * row_bytes))
            finally:
                cpp().close_parquet(h)
        # boundaries: accumulate estimated bytes in hi order; cut at ~1/4
        # of the budget so decode intermediates stay well inside it
        budget = max(1, limit // 4)
        bounds = []
        acc = 0
        for _, _, _, hi, est in sorted(spans, key=lambda s: (s[3], s[2])):
            acc += est
            if acc >= budget:
                bounds.append(hi)
                acc = 0
        warnings.warn(
            f"scan unit bucket={unit.bucket_id} exceeds LAKESOUL_MAX_UNIT_BYTES;"
            f" merging in {len(bounds) + 1} PK ranges (chunked spill merge)")
        parts = []
        prev = None
        for bound in bounds + [None]:
            rg_sets = {}
            for path, rg, lo, hi, _ in spans:
                if (bound is None or lo <= bound) and (prev is None or hi > prev):
                    rg_sets.setdefault(path, []).append(rg)
            if rg_sets:
                file_cols, present = [], []
                for path in files:
                    if path not in rg_sets:
                        continue
                    cols, pres = self._read_file_cpu(path, self.read_cols,
                                                     rg_subset=sorted(rg_sets[path]))
                    file_cols.append(cols)
                    present.append(pres)
                merged = merge_sorted_files(
                    file_cols, self.pk, self.merge_ops, self.cdc_column, present
                )
                batch = self._np_to_batch(merged, unit)
                pkc = batch.columns[pk0]
                if pkc.is_string:
                    # string PK: compare decoded values the same way the
                    # stats were decoded (utf-8/replace) so cut points
                    # and row filters share one total order
                    offs = pkc.offsets.numpy()
                    by = pkc.bytes_.numpy().tobytes()
                    vals = np.array(
                        [by[offs[i]:offs[i + 1]].decode("utf-8", "replace")
                         for i in range(batch.num_rows)], dtype=object)
                    m_np = np.ones(batch.num_rows, dtype=bool)
                    if prev is not None:
                        m_np &= vals > prev
                    if bound is not None:
                        m_np &= vals <= bound
                    mask = torch.from_numpy(m_np)
                else:
                    pkv = pkc.data
                    mask = torch.ones(batch.num_rows, dtype=torch.bool)
                    if prev is not None:
                        mask &= pkv > prev
                    if bound is not None:
                        mask &= pkv <= bound
                idx = torch.nonzero(mask, as_tuple=True)[0]
                if idx.numel():
                    part = batch.take(idx)
                    if self.device == "cuda":
                        part = part.to_device("cuda")
                    parts.append(part)
            prev = bound
        if not parts:
            return None
        return concat_batches(parts)

    def _read_unit_cpu(self, unit: ScanUnit) -> Optional[Batch]:
        file_cols: List[Dict[str, NpColumn]] = []
        present: List[set] = []
        for path in unit.files:
            cols, pres = self._read_file_cpu(path, self.read_cols)
            file_cols.append(cols)
            present.append(pres)
        needs_merge = bool(self.pk) and (
            len(file_cols) > 1 or self.cdc_column is not None or bool(self.merge_ops)
        )
        if needs_merge:
            merged = merge_sorted_files(
                file_cols, self.pk, self.merge_ops, self.cdc_column, present
            )
        else:
            if len(file_cols) == 1:
                merged = file_cols[0]
            else:
                from .merge_cpu import _concat_column

                merged = {
                    name: _concat_column([fc[name] for fc in file_cols])
                    for name in self.read_cols
                }
        return self._np_to_batch(merged, unit)

    @staticmethod
    def _np_col_to_column(dtype: str, npc: NpColumn) -> Column:
        """Convert one merged NpColumn into a Batch Column."""
        validity = (None if npc.validity is None
                    else torch.from_numpy(npc.validity))
        if dtype == "list<string>":
            d = cpp().split_len_prefixed(
                torch.from_numpy(np.ascontiguousarray(npc.bytes_)),
                torch.from_numpy(np.ascontiguousarray(npc.offsets,
                                                      dtype=np.int64)))
            return Column(dtype, offsets=d["row_offsets"], bytes_=d["bytes"],
                          elem_offsets=d["elem_offsets"], validity=validity)
        if dtype.startswith("list<"):
            es = np.dtype(_np_phys(dtype[5:-1])).itemsize
            offs = np.ascontiguousarray(npc.offsets, dtype=np.int64) // es
            vals = np.ascontiguousarray(npc.bytes_).view(_np_phys(dtype[5:-1]))
            return Column(dtype, data=torch.from_numpy(vals.copy()),
                          offsets=torch.from_numpy(offs), validity=validity)
        if npc.is_string:
            return Column(
                dtype,
                offsets=torch.from_numpy(
                    np.ascontiguousarray(npc.offsets, dtype=np.int32)),
                bytes_=torch.from_numpy(np.ascontiguousarray(npc.bytes_)),
                validity=validity)
        data = npc.data.astype(_np_target(dtype), copy=False)
        return Column(dtype, data=torch.from_numpy(np.ascontiguousarray(data)),
                      validity=validity)

    def _np_to_batch(self, merged: Dict[str, NpColumn], unit: ScanUnit) -> Batch:
        from .schema import map_params, struct_members

        cols: Dict[str, Column] = {}
        for f in self.eval_schema:
            if f.name in self.range_cols:
                cols[f.name] = self._range_value_column(f, unit, merged)
                continue
            sm = struct_members(f.dtype)
            mp = map_params(f.dtype)
            if sm is not None:
                # reassemble from the dotted leaves (merge ran per leaf;
                # row alignment is shared, so the struct validity is any
                # member's validity)
                kids, validity = {}, None
                for mn, mt in sm:
                    mc = self._np_col_to_column(mt, merged[f"{f.name}.{mn}"])
                    validity = mc.validity if validity is None else validity
                    mc.validity = None
                    kids[mn] = mc
                cols[f.name] = Column(f.dtype, validity=validity, children=kids)
                continue
            if mp is not None:
                kt, vt = mp
                kc = self._np_col_to_column(f"list<{kt}>",
                                            merged[f"{f.name}.key"])
                vc = self._np_col_to_column(f"list<{vt}>",
                                            merged[f"{f.name}.value"])
                validity = kc.validity
                kc.validity = vc.validity = None
                cols[f.name] = Column(f.dtype, validity=validity,
                                      children={"key": kc, "value": vc})
                continue
            cols[f.name] = self._np_col_to_column(f.dtype, merged[f.name])
        return Batch(self.eval_schema, cols)

    def _range_value_column(self, f, unit: ScanUnit, merged) -> Column:
        """Materialize a range-partition column from the partition_desc.

        Values are sentinel-decoded (NULL/''/','/'=' round trip —
        reference helpers/mod.rs:325); the NULL sentinel materializes as
        an all-null column."""
        n = len(next(iter(merged.values())))
        val, found = None, False
        for kv in unit.partition_desc.split(","):
            if "=" in kv and kv.split("=", 1)[0] == f.name:
                val = constants.decode_partition_value(kv.split("=", 1)[1])
                found = True
        is_null = found and val is None
        validity = (torch.zeros(n, dtype=torch.uint8) if is_null else None)
        if f.is_fixed_width:
            npdt = _np_target(f.dtype)
            arr = np.full(n, npdt(val) if (found and not is_null) else 0, dtype=npdt)
            return Column(f.dtype, data=torch.from_numpy(arr), validity=validity)
        enc = (val or "").encode()
        offs = np.arange(n + 1, dtype=np.int32) * len(enc)
        bys = np.frombuffer(enc * n, dtype=np.uint8).copy() if n else np.empty(0, np.uint8)
        return Column(f.dtype, offsets=torch.from_numpy(offs),
                      bytes_=torch.from_numpy(bys), validity=validity)

    def _read_file_cpu(self, path: str, names: Sequence[str],
                       rg_subset: Optional[Sequence[int]] = None) -> Dict[str, NpColumn]:
        h = cpp().open_parquet(path)
        try:
            meta = cpp().parquet_meta(h)
            file_cols = {c["name"]: i for i, c in enumerate(meta["columns"])}
            nrg = meta["num_row_groups"]
            rg_iter = list(rg_subset) if rg_subset is not None else list(range(nrg))
            total = (meta["num_rows"] if rg_subset is None else
                     sum(meta["row_groups"][rg]["num_rows"] for rg in rg_iter))
            out: Dict[str, NpColumn] = {}
            rc = []
            req = []
            for name in names:
                if name in file_cols:
                    for rg in rg_iter:
                        rc.append((rg, file_cols[name]))
                    req.append(name)
            chunks = cpp().read_chunks_cpu_batch(h, rc, 0) if rc else []
            ci = 0
            for name in req:
                f = self._field_for(name)
                if f.dtype == "list<string>":
                    # rides the byte-string machinery as the PLAIN
                    # parquet stream ([u32 len][bytes] per element) so the
                    # opaque whole-row gather keeps element boundaries;
                    # split_len_prefixed() parses it back at the end
                    offs_parts, bytes_parts, masks = [], [], []
                    any_null = False
                    for rg in rg_iter:
                        d = chunks[ci]
                        ci += 1
                        eoffs = d["offsets"].numpy().astype(np.int64)
                        by = d["bytes"].numpy()
                        lo_ = d["list_offsets"].numpy()
                        lens = np.diff(eoffs)
                        m = len(lens)
                        new_eoffs = np.zeros(m + 1, np.int64)
                        np.cumsum(lens + 4, out=new_eoffs[1:])
                        out_b = np.empty(int(new_eoffs[-1]), np.uint8)
                        if m:
                            pref = lens.astype("<u4").view(np.uint8).reshape(m, 4)
                            out_b[(new_eoffs[:-1, None] +
                                   np.arange(4)).reshape(-1)] = pref.reshape(-1)
                            nb_ = int(eoffs[-1])
                            if nb_:
                                dst = (np.repeat(new_eoffs[:-1] + 4, lens) +
                                       np.arange(nb_) - np.repeat(eoffs[:-1], lens))
                                out_b[dst] = by[:nb_]
                        offs_parts.append(new_eoffs[lo_])
                        bytes_parts.append(out_b)
                        lv = d["list_validity"].numpy()
                        nrow = len(offs_parts[-1]) - 1
                        if len(lv) and not lv.all():
                            any_null = True
                            masks.append(lv[:nrow])
                        else:
                            masks.append(np.ones(nrow, dtype=np.uint8))
                    validity = np.concatenate(masks) if any_null else None
                    total_rows = sum(len(o) - 1 for o in offs_parts)
                    offs = np.zeros(total_rows + 1, dtype=np.int64)
                    pos, base = 0, 0
                    for o in offs_parts:
                        nr = len(o) - 1
                        offs[pos + 1: pos + nr + 1] = o[1:] + base
                        base += int(o[-1]) if len(o) else 0
                        pos += nr
                    bys = (np.concatenate(bytes_parts) if bytes_parts
                           else np.empty(0, np.uint8))
                    out[name] = NpColumn(f.dtype, None, offs, bys, validity)
                    continue
                if f.dtype.startswith("list<"):
                    # list<T> rides the byte-string machinery downstream:
                    # offsets in BYTES over the raw element buffer (merge/
                    # gather treat the whole list value opaquely — UseLast
                    # whole-value semantics, reference merge/mod.rs:65-89)
                    es = np.dtype(_np_phys(f.dtype[5:-1])).itemsize
                    offs_parts, bytes_parts, masks = [], [], []
                    any_null = False
                    for rg in rg_iter:
                        d = chunks[ci]
                        ci += 1
                        offs_parts.append(d["list_offsets"].numpy() * es)
                        bytes_parts.append(d["data"].numpy())
                        lv = d["list_validity"].numpy()
                        nrow = len(offs_parts[-1]) - 1
                        if len(lv) and not lv.all():
                            any_null = True
                            masks.append(lv[:nrow])
                        else:
                            masks.append(np.ones(nrow, dtype=np.uint8))
                    validity = np.concatenate(masks) if any_null else None
                    total_rows = sum(len(o) - 1 for o in offs_parts)
                    offs = np.zeros(total_rows + 1, dtype=np.int64)
                    pos, base = 0, 0
                    for o in offs_parts:
                        nr = len(o) - 1
                        offs[pos + 1: pos + nr + 1] = o[1:].astype(np.int64) + base
                        base += int(o[-1]) if len(o) else 0
                        pos += nr
                    bys = (np.concatenate(bytes_parts) if bytes_parts
                           else np.empty(0, np.uint8))
                    out[name] = NpColumn(f.dtype, None, offs, bys, validity)
                    continue
                parts, offs_parts, bytes_parts, masks = [], [], [], []
                any_null = False
                for rg in rg_iter:
                    d = chunks[ci]
                    ci += 1
                    nv = d["num_values"]
                    if f.dtype in ("string", "binary"):
                        offs_parts.append(d["offsets"].numpy())
                        bytes_parts.append(d["bytes"].numpy())
                    else:
                        parts.append(d["data"].numpy().view(_np_phys(f.dtype)))
                    v = d["validity"].numpy()
                    if len(v):
                        any_null = True
                        masks.append(v)
                    else:
                        masks.append(np.ones(nv, dtype=np.uint8))
                validity = np.concatenate(masks) if any_null else None
                if f.dtype in ("string", "binary"):
                    total_rows = sum(len(o) - 1 for o in offs_parts)
                    offs = np.zeros(total_rows + 1, dtype=np.int64)
                    pos, base = 0, 0
                    for o in offs_parts:
                        n = len(o) - 1
                        offs[pos + 1 : pos + n + 1] = o[1:].astype(np.int64) + base
                        base += int(o[-1]) if len(o) else 0
                        pos += n
                    bys = (
                        np.concatenate(bytes_parts)
                        if bytes_parts
                        else np.empty(0, np.uint8)
                    )
                    out[name] = NpColumn(f.dtype, None, offs, bys, validity)
                else:
                    data = np.concatenate(parts) if parts else np.empty(0, _np_phys(f.dtype))
                    out[name] = NpColumn(f.dtype, data, None, None, validity)
            # schema evolution: missing columns become nulls
            for name in names:
                if name in out:
                    continue
                f = self._field_for(name)
                if f.dtype in ("string", "binary") or f.dtype.startswith("list<"):
                    out[name] = NpColumn(
                        f.dtype,
                        None,
                        np.zeros(total + 1, dtype=np.int64),
                        np.empty(0, np.uint8),
                        np.zeros(total, dtype=np.uint8),
                    )
                else:
                    out[name] = NpColumn(
                        f.dtype,
                        np.zeros(total, dtype=_np_phys(f.dtype)),
                        None,
                        None,
                        np.zeros(total, dtype=np.uint8),
                    )
            return out, set(req)
        finally:
            cpp().close_parquet(h)

    # ------------------------------------------------------------------ #

    def _apply_filters(self, batch: Batch) -> Batch:
        if self.filter_expr is None:
            return batch
        keep = self.filter_expr.evaluate(batch)
        idx = torch.nonzero(keep, as_tuple=True)[0]
        if idx.numel() == batch.num_rows:
            return batch
        return batch.take(idx)

    def prune_files_by_stats(self, files: List[str]) -> List[str]:
        """File-level min/max pruning (the reference's Parquet row-group
        statistics pushdown). For PK tables only PK-column predicates are
        safe to prune on — dropping a delta file on a non-PK predicate
        would resurrect stale rows in the merge."""
        if self.filter_expr is None:
            return files
        from ..ops import cpp

        out = []
        for path in files:
            try:
                h = cpp().open_parquet(self._localize([path])[0])
            except Exception:
                out.append(path)
                continue
            try:
                meta = cpp().parquet_meta(h)
                cols = meta["columns"]
                stats: Dict[str, tuple] = {}
                for ci, cinfo in enumerate(cols):
                    name = cinfo["name"]
                    if self.pk and name not in self.pk:
                        continue
                    dtype = cinfo["dtype"]
                    mn = mx = None
                    ok = True
                    for rg in meta["row_groups"]:
                        sd = rg["columns"][ci]
                        if "min" not in sd:
                            ok = False
                            break
                        lo = decode_stat(sd["min"], dtype)
                        hi = decode_stat(sd["max"], dtype)
                        mn = lo if mn is None else min(mn, lo)
                        mx = hi if mx is None else max(mx, hi)
                    if ok and mn is not None:
                        stats[name] = (mn, mx)
                if self.filter_expr.prune_stats(stats):
                    out.append(path)
            finally:
                cpp().close_parquet(h)
        return out
