"""GPU read path: host IO/decompress -> single H2D -> HIP decode -> GPU merge.

Pipeline per scan unit:
1. host (C++ thread pool, one call per unit): footer parse, page walk,
   zstd/snappy decompress, def-level RLE decode, and layout of every
   chunk payload into a handful of contiguous (pinned) buffers
   (_cpp.read_unit_raw / csrc/cpp/read_unit.h);
2. ONE async H2D per buffer (values / validity / dicts / runs);
3. HIP kernels materialize columns in HBM: PLAIN fixed-width columns are
   zero-copy views into the device buffer; dictionary chunks expand with
   the RLE bit-unpack kernel + fused dict-gather/null-scatter; nullable
   PLAIN columns scatter through validity;
4. GPU merge-on-read (merge_gpu) with the merge-path kernel.

String columns are host-assembled ((len,bytes) stream is serial) and ship
as (offsets int64, bytes) into HBM.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from ..ops import cpp, hip
from ..utils import timing
from .batch import Batch, Column, torch_dtype_for
from .schema import Schema

_TORCH_VIEW = {
    "bool": torch.uint8,
    "int8": torch.int32,   # physical INT32; narrowed after decode
    "int16": torch.int32,
    "int32": torch.int32,
    "int64": torch.int64,
    "float32": torch.float32,
    "float64": torch.float64,
    "date32": torch.int32,
    "timestamp[us]": torch.int64,
    "timestamp[ms]": torch.int64,
    "timestamp[ns]": torch.int64,
}
_TARGET = {"int8": torch.int8, "int16": torch.int16}


def _torch_view(dtype):
    return torch.int64 if dtype.startswith("decimal") else _TORCH_VIEW[dtype]
_ESIZE = {torch.uint8: 1, torch.int32: 4, torch.int64: 8, torch.float32: 4, torch.float64: 8}


import os as _os

def _host_threads_per_rank() -> int:
    try:
        quota_s = open("/sys/fs/cgroup/cpu.max").read().split()
        ncpu = _os.cpu_count() or 16
        quota = ncpu if quota_s[0] == "max" else max(
            1, int(quota_s[0]) // int(quota_s[1]))
    except OSError:
        quota = _os.cpu_count() or 16
    world = int(_os.environ.get("WORLD_SIZE", "1"))
    return max(1, quota // max(1, world))


def _default_gpu_zstd_frac() -> float:
    """Fraction of zstd chunks decoded by the GPU kernel (the rest go to
    the host pool; the prefetch pipeline overlaps the two across units).

    Balanced split from measured single-source rates (profiles/
    r01_gpu_zstd.md: host pool ~90 ms/step at 16 threads, GPU v2 kernel
    ~136 ms/step): frac = host_rate/(host_rate+gpu_rate) scaled by the
    rank's actual thread budget. Starved ranks (<3 threads) go full GPU;
    LAKESOUL_GPU_ZSTD=0/1 still forces pure host / pure GPU and
    LAKESOUL_GPU_ZSTD_FRAC pins an explicit split."""
    env = _os.environ.get("LAKESOUL_GPU_ZSTD")
    if env is not None and env != "":
        if env == "0":
            return 0.0
        if env == "1":
            return 1.0
    envf = _os.environ.get("LAKESOUL_GPU_ZSTD_FRAC")
    if envf:
        return min(1.0, max(0.0, float(envf)))
    threads = _host_threads_per_rank()
    # Measured r2 A/B on the headline bench (gpurun bench_frac_*.log):
    # host-only 40 ms/step, frac=0.4 342 ms, frac=1.0 371 ms at the
    # 128 KB default pages — the GPU kernel's wave-per-page parallelism
    # collapses on large pages (it wants thousands of 32 KB pages per
    # unit, profiles/r01_gpu_zstd.md). Hybrid stays opt-in via
    # LAKESOUL_GPU_ZSTD_FRAC (+ LAKESOUL_PAGE_BYTES=32768 at write time);
    # the adaptive rule keeps full-GPU decode for starved ranks only.
    if threads < 6:
        return 1.0
    return 0.0


_GPU_ZSTD_FRAC = _default_gpu_zstd_frac()
_GPU_ZSTD = _GPU_ZSTD_FRAC > 0.0


def fetch_raw(files: List[str], names: List[str]) -> dict:
    """Host phase of a unit read (releases the GIL in C++) — safe to run
    on a prefetch thread while the GPU processes the previous unit."""
    with timing.phase("host_fetch"):
        raw = cpp().read_unit_raw(files, names, 0, True, True, _GPU_ZSTD,
                                  _GPU_ZSTD_FRAC)
    if timing.ENABLED:
        timing._acc["fetch.stage1"] += raw["t_stage1_us"] / 1e6
        timing._acc["fetch.s1_open"] += raw["t_open_us"] / 1e6
        timing._acc["fetch.s1_chunks"] += raw["t_chunks_us"] / 1e6
        timing._acc["fetch.s1_layout"] += raw["t_layout_us"] / 1e6
        timing._acc["fetch.alloc"] += raw["t_alloc_us"] / 1e6
        timing._acc["fetch.fill"] += raw["t_fill_us"] / 1e6
        timing._cnt["fetch.stage1"] += 1
    return raw


_copy_stream: Optional["torch.cuda.Stream"] = None


def _get_copy_stream():
    global _copy_stream
    if _copy_stream is None:
        _copy_stream = torch.cuda.Stream()
    return _copy_stream


class UnitTransfer:
    """Async H2D of one unit's buffers (default stream: the python thread
    runs ahead of the GPU, so next-unit copies already pipeline behind
    the current unit's kernels; a dedicated copy stream measured ~8%
    SLOWER from event/ordering overhead — kept default)."""

    def __init__(self, raw: dict, device):
        self.raw = raw  # hold pinned host buffers until consumed
        # values regions covered by GPU decompress jobs carry no host
        # data — ship only the host-filled gaps (for all-zstd units the
        # H2D volume drops to the compressed bytes)
        jobs_host = []
        for k in ("snappy_jobs", "zstd_jobs"):
            t = raw.get(k)
            if t is not None and t.numel():
                jobs_host.append(t.view(-1, 4))
        nvals = raw["values"].numel()
        if jobs_host and nvals:
            import torch as _t

            import numpy as _np

            j = _t.cat(jobs_host) if len(jobs_host) > 1 else jobs_host[0]
            dst = j[:, 2].numpy()
            ln = j[:, 3].numpy()
            order = dst.argsort()
            dst, ln = dst[order], ln[order]
            ends = _np.maximum.accumulate(dst + ln)
            gap_starts = _np.concatenate(([0], ends))
            gap_ends = _np.concatenate((dst, [nvals]))
            keep = gap_ends > gap_starts
            self.vals = _t.empty(nvals, dtype=_t.uint8, device=device)
            hv = raw["values"]
            for a, b in zip(gap_starts[keep], gap_ends[keep]):
                self.vals[a:b].copy_(hv[a:b], non_blocking=True)
        else:
            self.vals = raw["values"].to(device, non_blocking=True)
        self.validity = (
            raw["validity"].to(device, non_blocking=True)
            if raw["validity"].numel() else None
        )
        self.dicts = raw["dicts"].to(device, non_blocking=True) if raw["dicts"].numel() else None
        self.runs = (
            raw["runs"].view(-1, 6).to(device, non_blocking=True)
            if raw["runs"].numel() else None
        )
        self.soffs = raw["soffs"].to(device, non_blocking=True) if raw["soffs"].numel() else None
        self.comp = raw["comp"].to(device, non_blocking=True) if raw.get("comp") is not None and raw["comp"].numel() else None
        self.snappy_jobs = (
            raw["snappy_jobs"].view(-1, 4).to(device, non_blocking=True)
            if raw.get("snappy_jobs") is not None and raw["snappy_jobs"].numel()
            else None
        )
        self.zstd_jobs = (
            raw["zstd_jobs"].view(-1, 4).to(device, non_blocking=True)
            if raw.get("zstd_jobs") is not None and raw["zstd_jobs"].numel()
            else None
        )
        self.event = None



def _project_eval(scan, merged, unit, device):
    """Project the merged (leaf-level) batch to scan.eval_schema:
    materialize range columns, convert byte-offset list lanes back to
    element-offset list columns, reassemble struct/map columns from
    their dotted leaves."""
    from .schema import map_params, struct_members

    def conv_list(lt, c):
        es = _ESIZE[_torch_view(lt[5:-1])]
        vals_t = (c.bytes_.view(torch_dtype_for(lt[5:-1]))
                  if c.bytes_.numel()
                  else torch.empty(0, dtype=torch_dtype_for(lt[5:-1]),
                                   device=device))
        return Column(lt, data=vals_t,
                      offsets=c.offsets.to(torch.int64) // es,
                      validity=c.validity)

    cols: Dict[str, Column] = {}
    nrows = merged.num_rows
    for f in scan.eval_schema:
        if f.name in scan.range_cols:
            cols[f.name] = _range_col_gpu(scan, f, unit, nrows, device)
            continue
        sm = struct_members(f.dtype)
        mp = map_params(f.dtype)
        if sm is not None:
            kids, validity = {}, None
            for mn, mt in sm:
                mc = merged.columns[f"{f.name}.{mn}"]
                if validity is None:
                    validity = mc.validity
                kids[mn] = Column(mt, data=mc.data, offsets=mc.offsets,
                                  bytes_=mc.bytes_,
                                  elem_offsets=mc.elem_offsets)
            cols[f.name] = Column(f.dtype, validity=validity, children=kids)
            continue
        if mp is not None:
            kt, vt = mp
            kc = conv_list(f"list<{kt}>", merged.columns[f"{f.name}.key"])
            vc = conv_list(f"list<{vt}>", merged.columns[f"{f.name}.value"])
            validity = kc.validity
            kc.validity = vc.validity = None
            cols[f.name] = Column(f.dtype, validity=validity,
                                  children={"key": kc, "value": vc})
            continue
        if f.dtype.startswith("list<"):
            cols[f.name] = conv_list(f.dtype, merged.columns[f.name])
            continue
        cols[f.name] = merged.columns[f.name]
    return Batch(scan.eval_schema, cols)


def read_unit_gpu(scan, unit, raw: Optional[dict] = None,
                  transfer: Optional[UnitTransfer] = None) -> Optional[Batch]:
    from .merge_gpu import merge_sorted_files_gpu

    device = torch.device("cuda")
    names = scan.read_cols
    if transfer is None:
        if raw is None:
            raw = fetch_raw(unit.files, names)
        with timing.phase("h2d"):
            transfer = UnitTransfer(raw, device)
    raw = transfer.raw
    if transfer.event is not None:
        torch.cuda.current_stream().wait_event(transfer.event)
    vals = transfer.vals
    validity_buf = transfer.validity
    dicts_buf = transfer.dicts
    runs_buf = transfer.runs
    soffs_buf = transfer.soffs
    if transfer.snappy_jobs is not None:
        # GPU snappy: decompress page bodies straight into the values
        # buffer (wave-per-page kernel) — no host decompression happened
        # for these chunks
        status = hip().snappy_decompress_into(transfer.comp, transfer.snappy_jobs, vals)
        if bool((status != 0).any()):
            raise RuntimeError(f"GPU snappy decompression failed: {status.cpu().tolist()}")
    if transfer.zstd_jobs is not None:
        # GPU zstd: the from-scratch RFC 8878 decoder (csrc/hip/zstd.hip)
        # decompresses zstd page frames straight into the values buffer —
        # compressed bytes crossed the bus, the cgroup-capped host CPUs
        # never touched them
        status = hip().zstd_decompress_into(transfer.comp, transfer.zstd_jobs, vals)
        if bool((status != 0).any()):
            raise RuntimeError(f"GPU zstd decompression failed: status={int((status != 0).sum())} pages")

    from .schema import Schema as _RSch

    read_schema = _RSch([scan._field_for(n) for n in names])
    # list<T> columns travel the unit as BYTE-offset binary (element
    # payload in the string-bytes region) so the merge machinery treats
    # whole lists opaquely (UseLast whole-value, merge/mod.rs:65-89);
    # converted back to element-offset list columns at the end
    has_list = any(f.dtype.startswith("list<") for f in read_schema)
    if has_list:
        from .schema import Field as _Fld
        from .schema import Schema as _Sch

        work_schema = _Sch([
            _Fld(f.name, "binary" if f.dtype.startswith("list<") else f.dtype,
                 f.nullable)
            for f in read_schema])
    else:
        work_schema = read_schema
    ncols = len(names)
    empty_u8 = torch.empty(0, dtype=torch.uint8, device=device)
    empty_i64 = torch.empty(0, dtype=torch.int64, device=device)

    # ---- C++ fast path: decode + UseLast merge + gather in ONE call ----
    # (the per-unit python op storm was the scan's critical path)
    nfiles = len(raw["file_rows"])
    desc = raw.get("desc")
    if (
        desc is not None
        and not has_list
        and nfiles > 1
        and len(scan.pk) == 1
        and not scan.merge_ops
        and scan.cdc_column is None
        and scan.pk[0] in names
        and scan.schema.field(scan.pk[0]).dtype in ("int64", "int32")
        and bool(desc[:, 0].all())
    ):
        with timing.phase("gpu_unit_cpp", sync_gpu=False):
            pk_ci = names.index(scan.pk[0])
            pk_es = 8 if scan.schema.field(scan.pk[0]).dtype == "int64" else 4
            outs = hip().scan_unit_uselast(
                vals,
                validity_buf if validity_buf is not None else empty_u8,
                dicts_buf if dicts_buf is not None else empty_u8,
                runs_buf if runs_buf is not None else empty_i64,
                soffs_buf if soffs_buf is not None else empty_i64,
                desc,
                nfiles,
                ncols,
                pk_ci,
                pk_es,
            )
        cols: Dict[str, Column] = {}
        for ci, name in enumerate(names):
            f = scan._field_for(name)
            entry = outs[ci]
            if f.dtype in ("string", "binary"):
                offs, by, vmask = entry[0], entry[1], entry[2]
                cols[name] = Column(f.dtype, offsets=offs, bytes_=by, validity=vmask)
            else:
                data, vmask = entry[0], entry[1]
                data = data.view(_torch_view(f.dtype))
                if f.dtype in _TARGET:
                    data = data.to(_TARGET[f.dtype])
                cols[name] = Column(f.dtype, data=data, validity=vmask)
        merged = Batch(read_schema, cols)
        return _project_eval(scan, merged, unit, device)

    file_batches: List[Batch] = []
    present: List[set] = []
    _dec = timing.phase("gpu_decode", sync_gpu=True)
    _dec.__enter__()
    for fi, nrows in enumerate(raw["file_rows"]):
        cols: Dict[str, Column] = {}
        pres = set()
        for ci, name in enumerate(names):
            cd = raw["cols"][fi * ncols + ci]
            f = scan._field_for(name)
            is_list_f = f.dtype.startswith("list<")
            if not cd["present"]:
                if f.dtype in ("string", "binary") or is_list_f:
                    cols[name] = Column(
                        "binary" if is_list_f else f.dtype,
                        offsets=torch.zeros(nrows + 1, dtype=torch.int64, device=device),
                        bytes_=empty_u8,
                        validity=torch.zeros(nrows, dtype=torch.uint8, device=device),
                    )
                else:
                    cols[name] = Column(
                        f.dtype,
                        data=torch.zeros(nrows, dtype=torch_dtype_for(f.dtype), device=device),
                        validity=torch.zeros(nrows, dtype=torch.uint8, device=device),
                    )
                continue
            pres.add(name)
            nv = cd["num_values"]
            vmask = None
            if cd["validity_off"] >= 0 and cd["null_count"] > 0:
                vmask = validity_buf.narrow(0, cd["validity_off"], nv)

            if cd.get("is_list"):
                es = _ESIZE[_torch_view(f.dtype[5:-1])]
                offs = soffs_buf.narrow(0, cd["soff_off"], nv + 1).to(torch.int64) * es
                by = vals.narrow(0, cd["sbytes_off"], cd["sbytes_len"])
                cols[name] = Column("binary", offsets=offs, bytes_=by,
                                    validity=vmask)
                continue
            if cd["is_string"]:
                offs = soffs_buf.narrow(0, cd["soff_off"], nv + 1)
                by = vals.narrow(0, cd["sbytes_off"], cd["sbytes_len"])
                cols[name] = Column(f.dtype, offsets=offs, bytes_=by, validity=vmask)
                continue

            tdt = _torch_view(f.dtype)
            esize = _ESIZE[tdt]
            if cd["is_dict"]:
                runs = runs_buf.narrow(0, cd["run_off"], cd["run_cnt"])
                payload = vals.narrow(0, cd["val_off"], cd["val_len"])
                idx = hip().rle_expand(payload, runs, cd["dense_n"])
                dict_vals = dicts_buf.narrow(0, cd["dict_off"], cd["dict_len"])
                if vmask is not None:
                    pos = torch.cumsum(vmask.to(torch.int64), 0) - 1
                    data = hip().dict_gather_scatter(dict_vals, idx, vmask, pos, esize, nv)
                else:
                    data = hip().dict_gather_scatter(
                        dict_vals, idx, empty_u8, empty_i64, esize, nv
                    )
                data = data.view(tdt)
            else:
                dense = vals.narrow(0, cd["val_off"], cd["val_len"])
                if vmask is not None:
                    pos = torch.cumsum(vmask.to(torch.int64), 0) - 1
                    data = hip().scatter_valid(dense, vmask, pos, esize, nv).view(tdt)
                else:
                    data = dense.view(tdt)
            if f.dtype in _TARGET:
                data = data.to(_TARGET[f.dtype])
            cols[name] = Column(f.dtype, data=data, validity=vmask)
        file_batches.append(Batch(work_schema, cols))
        present.append(pres)
    _dec.__exit__(None, None, None)

    needs_merge = bool(scan.pk) and (
        len(file_batches) > 1 or scan.cdc_column is not None or bool(scan.merge_ops)
    )
    if needs_merge:
        with timing.phase("gpu_merge", sync_gpu=True):
            merged = merge_sorted_files_gpu(
                file_batches, scan.pk, scan.merge_ops, scan.cdc_column, present
            )
    elif True:
        from .batch import concat_batches

        merged = concat_batches(file_batches)

    # project to eval schema (+ materialize range-partition columns);
    # filter-only columns are dropped after filter evaluation
    return _project_eval(scan, merged, unit, device)


def _decode_fixed_chunk_gpu(d: dict, dtype: str, device) -> Column:
    """Decode one raw column chunk (fixed width) — standalone helper used
    by kernel unit tests; the production path is read_unit_gpu."""
    tdt = _torch_view(dtype)
    esize = _ESIZE[tdt]
    nv = d["num_values"]
    has_nulls = d["validity"].numel() > 0 and d["null_count"] > 0
    validity = d["validity"].to(device) if d["validity"].numel() else None
    empty_u8 = torch.empty(0, dtype=torch.uint8, device=device)
    empty_i64 = torch.empty(0, dtype=torch.int64, device=device)
    if d["is_dict"]:
        payload, runs, dense_n = cpp().prep_rle_runs(d["values"], d["idx_pages"])
        idx = hip().rle_expand(payload.to(device), runs.to(device), dense_n)
        dict_vals = d["dict"].to(device)
        if has_nulls:
            pos = torch.cumsum(validity.to(torch.int64), 0) - 1
            raw = hip().dict_gather_scatter(dict_vals, idx, validity, pos, esize, nv)
        else:
            raw = hip().dict_gather_scatter(dict_vals, idx, empty_u8, empty_i64, esize, nv)
        data = raw.view(tdt)
    else:
        dense = d["values"].to(device)
        if has_nulls:
            pos = torch.cumsum(validity.to(torch.int64), 0) - 1
            data = hip().scatter_valid(dense, validity, pos, esize, nv).view(tdt)
        else:
            data = dense.view(tdt)
    if dtype in _TARGET:
        data = data.to(_TARGET[dtype])
    return Column(dtype, data=data, validity=validity if has_nulls else None)


def _range_col_gpu(scan, f, unit, n, device) -> Column:
    val = None
    for kv in unit.partition_desc.split(","):
        if "=" in kv and kv.split("=", 1)[0] == f.name:
            val = kv.split("=", 1)[1]
    if f.is_fixed_width:
        tdt = torch_dtype_for(f.dtype)
        x = float(val) if f.dtype.startswith("float") else int(val) if val is not None else 0
        return Column(f.dtype, data=torch.full((n,), x, dtype=tdt, device=device))
    enc = (val or "").encode()
    offs = torch.arange(n + 1, dtype=torch.int64, device=device) * len(enc)
    bys = (
        torch.frombuffer(bytearray(enc * n), dtype=torch.uint8).to(device)
        if n
        else torch.empty(0, dtype=torch.uint8, device=device)
    )
    return Column(f.dtype, offsets=offs, bytes_=bys)
