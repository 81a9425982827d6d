"""GPU read path: host IO/decompress -> H2D -> HIP decode -> GPU merge.

Pipeline per scan unit:
1. host (C++ threads): footer parse, page walk, zstd/snappy decompress,
   def-level RLE decode to validity bytes (_cpp.read_chunks_raw_batch);
2. H2D copy of PLAIN payloads / dict pages / index payloads;
3. HIP kernels: RLE/bit-unpack dictionary-index expansion, dict gather,
   validity scatter — columns materialize directly in HBM as torch
   tensors (zero further copies);
4. GPU merge-on-read (merge_gpu) with the merge-path kernel.

String columns use the host decoder for offset assembly (the (len,bytes)
stream is inherently serial) and ship (offsets,bytes) to HBM.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from ..ops import cpp, hip
from .batch import Batch, Column
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
_TARGET = {
    "int8": torch.int8,
    "int16": torch.int16,
}
_ESIZE = {torch.uint8: 1, torch.int32: 4, torch.int64: 8, torch.float32: 4, torch.float64: 8}


def _decode_fixed_chunk_gpu(d: dict, dtype: str, device) -> Column:
    """Decode one raw column chunk (fixed width) into an HBM tensor."""
    tdt = _TORCH_VIEW[dtype]
    esize = _ESIZE[tdt]
    nv = d["num_values"]
    null_count = d["null_count"]
    has_nulls = d["validity"].numel() > 0 and null_count > 0
    validity = d["validity"].to(device, non_blocking=True) if d["validity"].numel() else None

    if d["is_dict"]:
        payload, runs, dense_n = cpp().prep_rle_runs(d["values"], d["idx_pages"])
        idx = hip().rle_expand(
            payload.to(device, non_blocking=True), runs.to(device, non_blocking=True), dense_n
        )
        dict_vals = d["dict"].to(device, non_blocking=True)
        if has_nulls:
            pos = torch.cumsum(validity.to(torch.int64), 0) - 1
            raw = hip().dict_gather_scatter(dict_vals, idx, validity, pos, esize, nv)
        else:
            raw = hip().dict_gather_scatter(
                dict_vals, idx,
                torch.empty(0, dtype=torch.uint8, device=device),
                torch.empty(0, dtype=torch.int64, device=device), esize, nv,
            )
        data = raw.view(tdt)
    else:
        dense = d["values"].to(device, non_blocking=True)
        if has_nulls:
            pos = torch.cumsum(validity.to(torch.int64), 0) - 1
            raw = hip().scatter_valid(dense, validity, pos, esize, nv)
            data = raw.view(tdt)
        else:
            data = dense.view(tdt)
    if dtype in _TARGET:
        data = data.to(_TARGET[dtype])
    v = validity if (has_nulls and validity is not None) else None
    return Column(dtype, data=data, validity=v)


def read_file_gpu(scan, path: str, names: List[str], device) -> Tuple[Dict[str, Column], set]:
    h = cpp().open_parquet(path)
    try:
        meta = cpp().parquet_meta(h)
        file_cols = {c["name"]: i for i, c in enumerate(meta["columns"])}
        nrg = meta["num_row_groups"]
        total = meta["num_rows"]

        fixed_req = [n for n in names if n in file_cols and not scan.schema.field(n).dtype in ("string", "binary")]
        str_req = [n for n in names if n in file_cols and scan.schema.field(n).dtype in ("string", "binary")]

        rc = []
        for name in fixed_req:
            for rg in range(nrg):
                rc.append((rg, file_cols[name]))
        raw = cpp().read_chunks_raw_batch(h, rc, 0) if rc else []
        out: Dict[str, Column] = {}
        ci = 0
        for name in fixed_req:
            f = scan.schema.field(name)
            parts = []
            for rg in range(nrg):
                parts.append(_decode_fixed_chunk_gpu(raw[ci], f.dtype, device))
                ci += 1
            if len(parts) == 1:
                out[name] = parts[0]
            else:
                data = torch.cat([p.data for p in parts])
                validity = None
                if any(p.validity is not None for p in parts):
                    validity = torch.cat(
                        [
                            p.validity
                            if p.validity is not None
                            else torch.ones(p.data.numel(), dtype=torch.uint8, device=device)
                            for p in parts
                        ]
                    )
                out[name] = Column(f.dtype, data=data, validity=validity)

        # strings: host decode, ship to HBM
        src = []
        for name in str_req:
            for rg in range(nrg):
                src.append((rg, file_cols[name]))
        dec = cpp().read_chunks_cpu_batch(h, src, 0) if src else []
        ci = 0
        for name in str_req:
            f = scan.schema.field(name)
            offs_parts, bytes_parts, masks = [], [], []
            any_null = False
            for rg in range(nrg):
                d = dec[ci]
                ci += 1
                offs_parts.append(d["offsets"].numpy())
                bytes_parts.append(d["bytes"])
                v = d["validity"]
                if v.numel():
                    any_null = True
                    masks.append(v)
                else:
                    masks.append(torch.ones(d["num_values"], dtype=torch.uint8))
            trows = sum(len(o) - 1 for o in offs_parts)
            offs = np.zeros(trows + 1, dtype=np.int32)
            pos, base = 0, 0
            for o in offs_parts:
                k = len(o) - 1
                offs[pos + 1 : pos + k + 1] = o[1:] + base
                base += int(o[-1]) if len(o) else 0
                pos += k
            out[name] = Column(
                f.dtype,
                offsets=torch.from_numpy(offs).to(device),
                bytes_=torch.cat(bytes_parts).to(device) if bytes_parts else torch.empty(0, dtype=torch.uint8, device=device),
                validity=torch.cat(masks).to(device) if any_null else None,
            )

        present = set(fixed_req) | set(str_req)
        # schema evolution: null-fill missing
        for name in names:
            if name in out:
                continue
            f = scan.schema.field(name)
            if f.dtype in ("string", "binary"):
                out[name] = Column(
                    f.dtype,
                    offsets=torch.zeros(total + 1, dtype=torch.int32, device=device),
                    bytes_=torch.empty(0, dtype=torch.uint8, device=device),
                    validity=torch.zeros(total, dtype=torch.uint8, device=device),
                )
            else:
                from .batch import torch_dtype_for

                out[name] = Column(
                    f.dtype,
                    data=torch.zeros(total, dtype=torch_dtype_for(f.dtype), device=device),
                    validity=torch.zeros(total, dtype=torch.uint8, device=device),
                )
        return out, present
    finally:
        cpp().close_parquet(h)


def read_unit_gpu(scan, unit) -> Optional[Batch]:
    from .merge_gpu import merge_sorted_files_gpu

    device = torch.device("cuda")
    read_schema = scan.schema.select(scan.read_cols)
    file_batches: List[Batch] = []
    present: List[set] = []
    for path in unit.files:
        cols, pres = read_file_gpu(scan, path, scan.read_cols, device)
        file_batches.append(Batch(read_schema, cols))
        present.append(pres)

    needs_merge = bool(scan.pk) and (
        len(file_batches) > 1 or scan.cdc_column is not None or bool(scan.merge_ops)
    )
    if needs_merge:
        merged = merge_sorted_files_gpu(
            file_batches, scan.pk, scan.merge_ops, scan.cdc_column, present
        )
    else:
        from .batch import concat_batches

        merged = concat_batches(file_batches)

    # project to out_schema (+ materialize range-partition columns)
    cols: Dict[str, Column] = {}
    nrows = merged.num_rows
    for f in scan.out_schema:
        if f.name in scan.range_cols:
            cols[f.name] = _range_col_gpu(scan, f, unit, nrows, device)
        else:
            cols[f.name] = merged.columns[f.name]
    return Batch(scan.out_schema, cols)


def _range_col_gpu(scan, f, unit, n, device) -> Column:
    val = None
    for kv in unit.partition_desc.split(","):
        if "=" in kv and kv.split("=", 1)[0] == f.name:
            val = kv.split("=", 1)[1]
    if f.is_fixed_width:
        from .batch import torch_dtype_for

        tdt = torch_dtype_for(f.dtype)
        x = float(val) if f.dtype.startswith("float") else int(val) if val is not None else 0
        return Column(f.dtype, data=torch.full((n,), x, dtype=tdt, device=device))
    enc = (val or "").encode()
    offs = torch.arange(n + 1, dtype=torch.int32, device=device) * len(enc)
    bys = torch.frombuffer(bytearray(enc * n), dtype=torch.uint8).to(device) if n else torch.empty(0, dtype=torch.uint8, device=device)
    return Column(f.dtype, offsets=offs, bytes_=bys)
