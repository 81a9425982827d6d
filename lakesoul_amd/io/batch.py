"""Column batch container — the in-memory currency of the engine.

Fixed-width columns are torch tensors (CPU or HBM-resident on GPU);
string/binary columns are (offsets int32, bytes uint8) pairs; NULLs are a
uint8 validity mask. The GPU reader materializes these directly in HBM;
``to_arrow``/``from_any`` bridge to pyarrow/pandas/numpy at the edges
(replacing the reference's Arrow C Data FFI plane).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Sequence

import numpy as np
import torch

from .schema import FIXED_WIDTH_BYTES, Field as LsField, Schema

_TORCH_DTYPE = {
    "bool": torch.uint8,
    "int8": torch.int8,
    "int16": torch.int16,
    "int32": torch.int32,
    "int64": torch.int64,
    "float16": torch.float16,
    "float32": torch.float32,
    "float64": torch.float64,
    "date32": torch.int32,
    "timestamp[us]": torch.int64,
    "timestamp[ms]": torch.int64,
    "timestamp[ns]": torch.int64,
}

_NP_DTYPE = {
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


def torch_dtype_for(dtype: str):
    if dtype.startswith("decimal"):
        return torch.int64   # unscaled int64 backing
    return _TORCH_DTYPE[dtype]


def np_dtype_for(dtype: str):
    if dtype.startswith("decimal"):
        return np.int64
    return _NP_DTYPE[dtype]


@dataclass
class Column:
    dtype: str
    data: Optional[torch.Tensor] = None      # fixed-width values
    offsets: Optional[torch.Tensor] = None   # int32 [n+1], string/binary
    bytes_: Optional[torch.Tensor] = None    # uint8, string/binary
    validity: Optional[torch.Tensor] = None  # uint8 [n], 1=valid; None=all valid
    # list<string> only: element -> byte offsets (int64 m+1); then
    # `offsets` is row -> element and `bytes_` the element payload
    elem_offsets: Optional[torch.Tensor] = None
    # struct<...>: child Columns per member (members are scalar, carry no
    # validity of their own — the struct's `validity` governs the row).
    # map<K,V>: children {"key": list<K> Column, "value": list<V> Column}
    # sharing row structure (same offsets).
    children: Optional[Dict[str, "Column"]] = None

    @property
    def is_string(self) -> bool:
        return self.dtype in ("string", "binary")

    @property
    def is_struct(self) -> bool:
        return self.dtype.startswith("struct<")

    @property
    def is_map(self) -> bool:
        return self.dtype.startswith("map<")

    @property
    def is_list(self) -> bool:
        # list<T>: `offsets` (int64 n+1, element ranges) + `data`
        # (element values tensor) + per-row validity
        return self.dtype.startswith("list<")

    @property
    def is_list_str(self) -> bool:
        return self.dtype == "list<string>"

    @property
    def elem_dtype(self) -> str:
        return self.dtype[5:-1]

    def __len__(self) -> int:
        if self.is_struct:
            return len(next(iter(self.children.values())))
        if self.is_string or self.is_list or self.is_map:
            return int(self.offsets.numel()) - 1 if self.offsets is not None \
                else len(self.children["key"])
        return int(self.data.numel())

    def to_numpy(self):
        if self.is_string:
            raise TypeError("string column: use offsets/bytes")
        return self.data.cpu().numpy()

    def to_device(self, device) -> "Column":
        # async only when moving TO the GPU: a non_blocking D2H copy into
        # pageable host memory returns before the bytes land, and callers
        # read the numpy view immediately (this was a real corruption —
        # batches read stale offsets when earlier GPU work kept the copy
        # queue busy; AMD_SERIALIZE_COPY=3 made it vanish)
        nb = torch.device(device).type == "cuda"

        def mv(t):
            return None if t is None else t.to(device, non_blocking=nb)

        kids = (None if self.children is None else
                {k: v.to_device(device) for k, v in self.children.items()})
        return Column(self.dtype, mv(self.data), mv(self.offsets), mv(self.bytes_),
                      mv(self.validity), mv(self.elem_offsets), kids)

    def take(self, idx: torch.Tensor) -> "Column":
        """Gather rows by index tensor (moved to this column's device —
        a CPU index against a CUDA column would reach the string gather
        kernel as a host pointer)."""
        if self.is_struct or self.is_map:
            kids = {k: v.take(idx) for k, v in self.children.items()}
            v = None if self.validity is None else self.validity[idx.to(
                self.validity.device)]
            return Column(self.dtype, validity=v, children=kids)
        dev = self.offsets.device if (self.is_string or self.is_list) else self.data.device
        if idx.device != dev:
            idx = idx.to(dev)
        if self.is_list_str:
            # two gathers: per-row payload bytes (string path over the
            # row's contiguous byte range) + per-row element byte lens
            # (numeric-list path over elem_offsets diffs), then cumsum
            # rebuilds element offsets in the new row order
            row_boffs = self.elem_offsets[self.offsets]
            byte_col = Column("binary", offsets=row_boffs, bytes_=self.bytes_,
                              validity=self.validity)
            got = byte_col.take(idx)
            elem_lens = self.elem_offsets[1:] - self.elem_offsets[:-1]
            lens_col = Column("list<int64>", data=elem_lens, offsets=self.offsets)
            lg = lens_col.take(idx)
            new_eoffs = torch.zeros(lg.data.numel() + 1, dtype=torch.int64,
                                    device=dev)
            torch.cumsum(lg.data, 0, out=new_eoffs[1:].view(-1))
            return Column(self.dtype, offsets=lg.offsets.to(torch.int64),
                          bytes_=got.bytes_, validity=got.validity,
                          elem_offsets=new_eoffs)
        if self.is_list:
            # gather element ranges through the byte-view of the values
            # (the string gather path, offsets scaled by element size)
            es = self.data.element_size()
            byte_col = Column(
                "binary",
                offsets=self.offsets.to(torch.int64) * es,
                bytes_=self.data.contiguous().view(torch.uint8)
                if self.data.numel() else torch.empty(0, dtype=torch.uint8,
                                                      device=dev),
                validity=self.validity)
            got = byte_col.take(idx)
            new_offs = got.offsets.to(torch.int64) // es
            vals = (got.bytes_.view(self.data.dtype) if got.bytes_.numel()
                    else torch.empty(0, dtype=self.data.dtype, device=dev))
            return Column(self.dtype, data=vals, offsets=new_offs,
                          validity=got.validity)
        if self.is_string:
            offs = self.offsets
            lens = offs[1:] - offs[:-1]
            new_lens = lens[idx]
            new_offs = torch.zeros(idx.numel() + 1, dtype=offs.dtype, device=offs.device)
            torch.cumsum(new_lens, 0, out=new_offs[1:].view(-1))
            # gather bytes (CPU loop-free path via numpy for now)
            if offs.device.type == "cpu":
                src_off = offs.numpy()
                src_b = self.bytes_.numpy()
                sel = idx.numpy()
                out = np.empty(int(new_offs[-1]), dtype=np.uint8)
                no = new_offs.numpy()
                for i, si in enumerate(sel):
                    out[no[i]:no[i + 1]] = src_b[src_off[si]:src_off[si + 1]]
                nb = torch.from_numpy(out)
            else:
                from ..ops import hip

                nb = hip().gather_strings(self.bytes_, self.offsets.to(torch.int64), idx.to(torch.int64), new_offs.to(torch.int64))
            v = None if self.validity is None else self.validity[idx]
            return Column(self.dtype, None, new_offs, nb, v)
        v = None if self.validity is None else self.validity[idx]
        return Column(self.dtype, self.data[idx], None, None, v)


def _slice_col(c: Column, a: int, b: int) -> Column:
    """Zero-copy row slice [a, b) of one column (views; variable-width
    columns rebase offsets; struct/map recurse into children)."""
    v = None if c.validity is None else c.validity[a:b]
    if c.is_struct or c.is_map:
        return Column(c.dtype, validity=v,
                      children={k: _slice_col(ch, a, b)
                                for k, ch in c.children.items()})
    if c.is_list_str:
        offs = c.offsets[a : b + 1]
        base = offs[0]
        lo, hi = int(base), int(c.offsets[b])
        eoffs = c.elem_offsets[lo : hi + 1]
        eb = int(eoffs[0]) if eoffs.numel() else 0
        return Column(
            c.dtype, offsets=(offs - base).to(torch.int64),
            bytes_=c.bytes_[eb:int(c.elem_offsets[hi])] if eoffs.numel()
            else c.bytes_[:0],
            validity=v, elem_offsets=(eoffs - eb).to(torch.int64))
    if c.is_list:
        offs = c.offsets[a : b + 1]
        base = offs[0]
        lo, hi = int(base), int(c.offsets[b])
        return Column(c.dtype, data=c.data[lo:hi],
                      offsets=(offs - base).to(c.offsets.dtype),
                      validity=v)
    if c.is_string:
        offs = c.offsets[a : b + 1]
        base = offs[0]
        lo = int(base)
        hi = int(c.offsets[b])
        return Column(c.dtype, None, (offs - base).to(c.offsets.dtype),
                      c.bytes_[lo:hi], v)
    return Column(c.dtype, c.data[a:b], None, None, v)


@dataclass
class Batch:
    schema: Schema
    columns: Dict[str, Column] = field(default_factory=dict)

    @property
    def num_rows(self) -> int:
        for c in self.columns.values():
            return len(c)
        return 0

    def column(self, name: str) -> Column:
        return self.columns[name]

    def to_device(self, device) -> "Batch":
        return Batch(self.schema, {k: v.to_device(device) for k, v in self.columns.items()})

    def take(self, idx: torch.Tensor) -> "Batch":
        return Batch(self.schema, {k: v.take(idx) for k, v in self.columns.items()})

    def slice(self, a: int, b: int) -> "Batch":
        """Zero-copy row slice [a, b) (views; strings rebase offsets)."""
        return Batch(self.schema,
                     {k: _slice_col(c, a, b) for k, c in self.columns.items()})

    # ------------------------------------------------------------------ #

    @classmethod
    def from_any(cls, data, schema: Schema) -> "Batch":
        """Build from pyarrow Table/RecordBatch, pandas DataFrame, or a
        dict of numpy arrays / torch tensors / lists."""
        mod = type(data).__module__
        if mod.startswith("pyarrow"):
            return cls.from_arrow(data, schema)
        if mod.startswith("pandas"):
            # nullable extension dtypes (Int32, boolean, string, …) carry an
            # NA mask that plain .to_numpy() destroys (fills with a garbage
            # sentinel); extract it as a validity mask. Plain float NaN stays
            # NaN (not null) — unchanged semantics.
            import pandas as _pd

            d, masks = {}, {}
            for c in data.columns:
                s = data[c]
                if _pd.api.types.is_extension_array_dtype(s.dtype):
                    na = s.isna().to_numpy()
                    if na.any():
                        masks[c] = (~na).astype(np.uint8)
                        fname_f = schema.field(c) if c in schema.names() else None
                        if fname_f is not None and not fname_f.is_fixed_width:
                            d[c] = [None if na[i] else s.iloc[i] for i in range(len(s))]
                        else:
                            d[c] = np.where(na, 0, s.to_numpy(dtype="object")).astype(
                                np_dtype_for(fname_f.dtype) if fname_f is not None else "int64")
                        continue
                    d[c] = s.to_numpy(
                        dtype=None if not c in schema.names() or not schema.field(c).is_fixed_width
                        else np_dtype_for(schema.field(c).dtype))
                else:
                    d[c] = s.to_numpy()
            b = cls.from_dict(d, schema)
            for c, m in masks.items():
                if b.columns[c].validity is None:
                    b.columns[c].validity = torch.from_numpy(m)
            return b
        if isinstance(data, dict):
            return cls.from_dict(data, schema)
        raise TypeError(f"unsupported data type {type(data)}")

    @classmethod
    def from_dict(cls, d: dict, schema: Schema) -> "Batch":
        cols = {}
        for f in schema:
            if f.name not in d:
                raise KeyError(f"missing column {f.name}")
            v = d[f.name]
            if f.dtype.startswith("struct<"):
                from .schema import struct_members

                members = struct_members(f.dtype)
                if isinstance(v, tuple) and len(v) in (1, 2):
                    kids = dict(v[0])
                    val = (torch.as_tensor(v[1], dtype=torch.uint8)
                           if len(v) == 2 and v[1] is not None else None)
                else:
                    items = list(v)
                    val = None
                    if any(x is None for x in items):
                        val = torch.tensor(
                            [0 if x is None else 1 for x in items],
                            dtype=torch.uint8)
                    child_schema = Schema([LsField(n, t, False)
                                           for n, t in members])
                    per = {}
                    for n, t in members:
                        dflt = "" if t in ("string", "binary") else 0
                        per[n] = [dflt if x is None else x[n] for x in items]
                    kids = cls.from_dict(per, child_schema).columns
                cols[f.name] = Column(f.dtype, validity=val, children=kids)
                continue
            if f.dtype.startswith("map<"):
                from .schema import map_params

                kt, vt = map_params(f.dtype)
                if isinstance(v, tuple) and len(v) in (1, 2):
                    kids = dict(v[0])
                    val = (torch.as_tensor(v[1], dtype=torch.uint8)
                           if len(v) == 2 and v[1] is not None else None)
                else:
                    items = list(v)
                    val = None
                    if any(x is None for x in items):
                        val = torch.tensor(
                            [0 if x is None else 1 for x in items],
                            dtype=torch.uint8)

                    def pairs(x):
                        if x is None:
                            return []
                        return list(x.items()) if isinstance(x, dict) else list(x)

                    keys = [[p[0] for p in pairs(x)] for x in items]
                    vals_ = [[p[1] for p in pairs(x)] for x in items]
                    kv_schema = Schema([LsField("key", f"list<{kt}>", False),
                                        LsField("value", f"list<{vt}>", False)])
                    kids = cls.from_dict({"key": keys, "value": vals_},
                                         kv_schema).columns
                cols[f.name] = Column(f.dtype, validity=val, children=kids)
                continue
            if f.dtype == "list<string>":
                if isinstance(v, tuple) and len(v) in (3, 4):
                    offs = torch.as_tensor(v[0], dtype=torch.int64)
                    eoffs = torch.as_tensor(v[1], dtype=torch.int64)
                    bys = torch.as_tensor(v[2], dtype=torch.uint8)
                    val = (torch.as_tensor(v[3], dtype=torch.uint8)
                           if len(v) == 4 and v[3] is not None else None)
                else:
                    items = list(v)
                    val = None
                    if any(x is None for x in items):
                        val = torch.tensor(
                            [0 if x is None else 1 for x in items],
                            dtype=torch.uint8)
                    offs = torch.zeros(len(items) + 1, dtype=torch.int64)
                    torch.cumsum(torch.tensor(
                        [0 if x is None else len(x) for x in items],
                        dtype=torch.int64), 0, out=offs[1:].view(-1))
                    enc = [s.encode() if isinstance(s, str) else bytes(s)
                           for x in items if x is not None for s in x]
                    eoffs = torch.zeros(len(enc) + 1, dtype=torch.int64)
                    torch.cumsum(torch.tensor([len(e) for e in enc],
                                              dtype=torch.int64), 0,
                                 out=eoffs[1:].view(-1))
                    bys = torch.from_numpy(np.frombuffer(
                        b"".join(enc), dtype=np.uint8).copy()) if enc else \
                        torch.empty(0, dtype=torch.uint8)
                cols[f.name] = Column(f.dtype, offsets=offs, bytes_=bys,
                                      validity=val, elem_offsets=eoffs)
                continue
            if f.dtype.startswith("list<"):
                elem_dt = f.dtype[5:-1]
                npdt = np_dtype_for(elem_dt)
                if isinstance(v, tuple) and len(v) in (2, 3):
                    offs = torch.as_tensor(v[0], dtype=torch.int64)
                    vals = torch.as_tensor(np.asarray(v[1], dtype=npdt))
                    val = (torch.as_tensor(v[2], dtype=torch.uint8)
                           if len(v) == 3 and v[2] is not None else None)
                else:
                    items = list(v)
                    validity_l = None
                    if any(x is None for x in items):
                        validity_l = torch.tensor(
                            [0 if x is None else 1 for x in items],
                            dtype=torch.uint8)
                    arrs = [np.asarray([] if x is None else x, dtype=npdt)
                            for x in items]
                    offs = torch.zeros(len(arrs) + 1, dtype=torch.int64)
                    torch.cumsum(torch.tensor([len(a) for a in arrs],
                                              dtype=torch.int64), 0,
                                 out=offs[1:].view(-1))
                    vals = torch.from_numpy(
                        np.concatenate(arrs) if arrs else np.empty(0, npdt))
                    val = validity_l
                cols[f.name] = Column(f.dtype, data=vals.to(
                    torch_dtype_for(elem_dt)), offsets=offs, validity=val)
                continue
            if f.is_fixed_width:
                if isinstance(v, torch.Tensor):
                    t = v
                elif isinstance(v, np.ndarray):
                    if v.dtype == np.bool_:
                        v = v.astype(np.uint8)
                    t = torch.from_numpy(np.ascontiguousarray(v))
                else:
                    t = torch.from_numpy(np.asarray(v, dtype=np_dtype_for(f.dtype)))
                t = t.to(torch_dtype_for(f.dtype))
                cols[f.name] = Column(f.dtype, data=t)
            else:
                if isinstance(v, tuple) and len(v) in (2, 3):
                    offs, by = v[0], v[1]
                    val = v[2] if len(v) == 3 else None
                    cols[f.name] = Column(
                        f.dtype,
                        offsets=torch.as_tensor(offs, dtype=torch.int32),
                        bytes_=torch.as_tensor(by, dtype=torch.uint8),
                        validity=None if val is None else torch.as_tensor(val, dtype=torch.uint8),
                    )
                else:
                    # list of str/bytes/None
                    items = list(v)
                    validity = None
                    if any(x is None for x in items):
                        validity = torch.tensor(
                            [0 if x is None else 1 for x in items], dtype=torch.uint8
                        )
                    enc = [
                        b"" if x is None else (x.encode() if isinstance(x, str) else bytes(x))
                        for x in items
                    ]
                    offs = np.zeros(len(enc) + 1, dtype=np.int32)
                    offs[1:] = np.cumsum([len(e) for e in enc])
                    cols[f.name] = Column(
                        f.dtype,
                        offsets=torch.from_numpy(offs),
                        bytes_=torch.from_numpy(
                            np.frombuffer(b"".join(enc), dtype=np.uint8).copy()
                        ),
                        validity=validity,
                    )
        return cls(schema, cols)

    @classmethod
    def from_arrow(cls, t, schema: Schema) -> "Batch":
        import pyarrow as pa

        if isinstance(t, pa.RecordBatch):
            t = pa.Table.from_batches([t])
        d = {}
        for f in schema:
            col = t.column(f.name)
            arr = col.combine_chunks() if hasattr(col, "combine_chunks") else col
            if f.dtype.startswith("decimal"):
                from .schema import decimal_params

                _, sc = decimal_params(f.dtype)
                import decimal as _dec

                vals = arr.to_pylist()
                validity = None
                if arr.null_count:
                    validity = torch.tensor(
                        [0 if x is None else 1 for x in vals], dtype=torch.uint8)
                q = _dec.Decimal(1).scaleb(-sc)
                unscaled = np.asarray(
                    [0 if x is None else int(x.quantize(q).scaleb(sc)) for x in vals],
                    dtype=np.int64,
                )
                c = Column(f.dtype, data=torch.from_numpy(unscaled), validity=validity)
                d[f.name] = c
                continue
            if f.is_fixed_width:
                np_arr = arr.to_numpy(zero_copy_only=False)
                validity = None
                if arr.null_count:
                    validity = (~np.asarray(arr.is_null())).astype(np.uint8)
                    np_arr = np.nan_to_num(np_arr) if np_arr.dtype.kind == "f" else np_arr
                    if np_arr.dtype == object:
                        np_arr = np.where(validity, np_arr, 0).astype(np_dtype_for(f.dtype))
                if np_arr.dtype == object or (np_arr.dtype.kind == "f" and f.dtype.startswith("int")):
                    np_arr = np.asarray(
                        [0 if x is None else x for x in arr.to_pylist()],
                        dtype=np_dtype_for(f.dtype),
                    )
                b = cls.from_dict({f.name: np_arr.astype(np_dtype_for(f.dtype))}, Schema([f]))
                c = b.columns[f.name]
                if validity is not None:
                    c.validity = torch.from_numpy(validity)
                d[f.name] = c
            else:
                d[f.name] = None  # placeholder, handled below
                items = arr.to_pylist()
                b = cls.from_dict({f.name: items}, Schema([f]))
                d[f.name] = b.columns[f.name]
        return cls(schema, d)

    def to_arrow(self):
        import pyarrow as pa

        from .schema import dtype_to_arrow

        arrays = []
        for f in self.schema:
            c = self.columns[f.name]
            if f.dtype.startswith("struct<"):
                from .schema import struct_members

                members = struct_members(f.dtype)
                child_b = Batch(
                    Schema([LsField(n, t, False) for n, t in members]),
                    c.children)
                ct = child_b.to_arrow()
                mask = None
                if c.validity is not None:
                    mask = pa.array(
                        ~c.validity.cpu().numpy().astype(bool))
                arrays.append(pa.StructArray.from_arrays(
                    [ct.column(n).combine_chunks() for n, _ in members],
                    [n for n, _ in members], mask=mask))
                continue
            if f.dtype.startswith("map<"):
                from .schema import map_params

                kt, vt = map_params(f.dtype)
                kv_b = Batch(
                    Schema([LsField("key", f"list<{kt}>", False),
                            LsField("value", f"list<{vt}>", False)]),
                    c.children)
                kvt = kv_b.to_arrow()
                karr = kvt.column("key").combine_chunks()
                varr = kvt.column("value").combine_chunks()
                n = len(c)
                offs_np = (c.children["key"].offsets.cpu().numpy()
                           .astype(np.int32, copy=False))
                mt = dtype_to_arrow(f.dtype)
                entries = pa.StructArray.from_arrays(
                    [karr.values, varr.values],
                    fields=[mt.key_field, mt.item_field])
                validity_buf = None
                null_count = 0
                if c.validity is not None:
                    vv = c.validity.cpu().numpy().astype(bool)
                    null_count = int(n - vv.sum())
                    validity_buf = pa.py_buffer(
                        np.packbits(vv, bitorder="little").tobytes())
                arrays.append(pa.Array.from_buffers(
                    dtype_to_arrow(f.dtype), n,
                    [validity_buf, pa.py_buffer(offs_np.tobytes())],
                    null_count=null_count, children=[entries]))
                continue
            if f.dtype == "list<string>":
                offs_np = c.offsets.cpu().numpy().astype(np.int32, copy=False)
                eoffs_np = c.elem_offsets.cpu().numpy().astype(np.int32, copy=False)
                m = len(eoffs_np) - 1
                vals = pa.Array.from_buffers(
                    pa.string(), m,
                    [None, pa.py_buffer(eoffs_np.tobytes()),
                     pa.py_buffer(c.bytes_.cpu().numpy().tobytes())])
                n = len(c)
                validity_buf = None
                null_count = 0
                if c.validity is not None:
                    vv = c.validity.cpu().numpy().astype(bool)
                    null_count = int(n - vv.sum())
                    validity_buf = pa.py_buffer(
                        np.packbits(vv, bitorder="little").tobytes())
                arrays.append(pa.Array.from_buffers(
                    dtype_to_arrow(f.dtype), n,
                    [validity_buf, pa.py_buffer(offs_np.tobytes())],
                    null_count=null_count, children=[vals]))
                continue
            if f.dtype.startswith("list<"):
                elem_dt = f.dtype[5:-1]
                offs_np = c.offsets.cpu().numpy().astype(np.int32, copy=False)
                vals = pa.array(c.data.cpu().numpy(),
                                type=dtype_to_arrow(elem_dt))
                n = len(c)
                validity_buf = None
                null_count = 0
                if c.validity is not None:
                    vv = c.validity.cpu().numpy().astype(bool)
                    null_count = int(n - vv.sum())
                    validity_buf = pa.py_buffer(
                        np.packbits(vv, bitorder="little").tobytes())
                arrays.append(pa.ListArray.from_arrays(
                    pa.Array.from_buffers(pa.int32(), n + 1,
                                          [None, pa.py_buffer(offs_np.tobytes())]),
                    vals) if validity_buf is None else
                    pa.Array.from_buffers(
                        dtype_to_arrow(f.dtype), n,
                        [validity_buf, pa.py_buffer(offs_np.tobytes())],
                        null_count=null_count, children=[vals]))
                continue
            if f.dtype.startswith("decimal"):
                v = c.data.cpu().numpy().astype(np.int64)
                n = v.size
                wide = np.empty((n, 2), dtype=np.int64)   # little-endian lo, hi
                wide[:, 0] = v
                wide[:, 1] = v >> 63                      # sign extension
                validity_buf = None
                null_count = 0
                if c.validity is not None:
                    vv = c.validity.cpu().numpy().astype(bool)
                    null_count = int(n - vv.sum())
                    validity_buf = pa.py_buffer(np.packbits(vv, bitorder="little").tobytes())
                arrays.append(pa.Array.from_buffers(
                    dtype_to_arrow(f.dtype), n,
                    [validity_buf, pa.py_buffer(wide.tobytes())],
                    null_count=null_count,
                ))
                continue
            if f.is_fixed_width:
                np_arr = c.data.cpu().numpy()
                if f.dtype == "bool":
                    np_arr = np_arr.astype(bool)
                mask = None
                if c.validity is not None:
                    mask = ~(c.validity.cpu().numpy().astype(bool))
                arrays.append(pa.array(np_arr, type=dtype_to_arrow(f.dtype), from_pandas=False, mask=mask))
            else:
                # zero-copy-ish buffer construction (offsets may be i64
                # from the GPU path; arrow string/binary wants i32)
                offs_np = c.offsets.cpu().numpy().astype(np.int32, copy=False)
                bys_np = c.bytes_.cpu().numpy()
                n = len(c)
                validity_buf = None
                null_count = 0
                if c.validity is not None:
                    v = c.validity.cpu().numpy().astype(bool)
                    null_count = int(n - v.sum())
                    validity_buf = pa.py_buffer(np.packbits(v, bitorder="little").tobytes())
                arr = pa.Array.from_buffers(
                    dtype_to_arrow(f.dtype),
                    n,
                    [validity_buf, pa.py_buffer(offs_np.tobytes()), pa.py_buffer(bys_np.tobytes())],
                    null_count=null_count,
                )
                arrays.append(arr)
        return pa.table(dict(zip(self.schema.names(), arrays)))


def concat_batches(batches: List[Batch]) -> Batch:
    if len(batches) == 1:
        return batches[0]
    schema = batches[0].schema
    cols = {}
    for f in schema:
        cs = [b.columns[f.name] for b in batches]
        n_total = sum(len(c) for c in cs)
        if cs and (cs[0].is_struct or cs[0].is_map):
            if cs[0].is_struct:
                from .schema import struct_members

                kid_fields = [LsField(n, t, False)
                              for n, t in struct_members(f.dtype)]
            else:
                from .schema import map_params

                kt, vt = map_params(f.dtype)
                kid_fields = [LsField("key", f"list<{kt}>", False),
                              LsField("value", f"list<{vt}>", False)]
            ks = Schema(kid_fields)
            kids = concat_batches(
                [Batch(ks, c.children) for c in cs]).columns
            validity = None
            if any(c.validity is not None for c in cs):
                dev0 = next(iter(kids.values()))
                dev0 = (dev0.offsets.device if dev0.offsets is not None
                        else dev0.data.device)
                validity = torch.cat([
                    c.validity if c.validity is not None
                    else torch.ones(len(c), dtype=torch.uint8, device=dev0)
                    for c in cs])
            cols[f.name] = Column(f.dtype, validity=validity, children=kids)
            continue
        if cs and cs[0].is_list_str:
            bytes_ = torch.cat([c.bytes_ for c in cs])
            m_total = sum(int(c.offsets[-1]) for c in cs)
            offs = torch.zeros(n_total + 1, dtype=torch.int64, device=bytes_.device)
            eoffs = torch.zeros(m_total + 1, dtype=torch.int64, device=bytes_.device)
            pos, ebase, epos, bbase = 0, 0, 0, 0
            for c in cs:
                n = len(c)
                m = int(c.offsets[-1])
                offs[pos + 1: pos + n + 1] = c.offsets[1:].to(torch.int64) + ebase
                eoffs[epos + 1: epos + m + 1] = c.elem_offsets[1:].to(torch.int64) + bbase
                ebase += m
                bbase += int(c.elem_offsets[-1])
                pos += n
                epos += m
            validity = None
            if any(c.validity is not None for c in cs):
                validity = torch.cat([
                    c.validity if c.validity is not None
                    else torch.ones(len(c), dtype=torch.uint8, device=bytes_.device)
                    for c in cs])
            cols[f.name] = Column(f.dtype, offsets=offs, bytes_=bytes_,
                                  validity=validity, elem_offsets=eoffs)
            continue
        if cs and cs[0].is_list:
            data = torch.cat([c.data for c in cs])
            offs = torch.zeros(n_total + 1, dtype=torch.int64, device=data.device)
            pos, base = 0, 0
            for c in cs:
                n = len(c)
                offs[pos + 1: pos + n + 1] = c.offsets[1:].to(torch.int64) + base
                base += int(c.offsets[-1])
                pos += n
            validity = None
            if any(c.validity is not None for c in cs):
                validity = torch.cat([
                    c.validity if c.validity is not None
                    else torch.ones(len(c), dtype=torch.uint8, device=data.device)
                    for c in cs])
            cols[f.name] = Column(f.dtype, data=data, offsets=offs,
                                  validity=validity)
            continue
        if f.is_fixed_width:
            data = torch.cat([c.data for c in cs])
            validity = None
            if any(c.validity is not None for c in cs):
                validity = torch.cat(
                    [
                        c.validity
                        if c.validity is not None
                        else torch.ones(len(c), dtype=torch.uint8, device=data.device)
                        for c in cs
                    ]
                )
            cols[f.name] = Column(f.dtype, data=data, validity=validity)
        else:
            bytes_ = torch.cat([c.bytes_ for c in cs])
            odt = cs[0].offsets.dtype
            offs = torch.zeros(n_total + 1, dtype=odt, device=bytes_.device)
            pos = 0
            base = 0
            for c in cs:
                n = len(c)
                offs[pos + 1 : pos + n + 1] = c.offsets[1:] + base
                base += int(c.offsets[-1])
                pos += n
            validity = None
            if any(c.validity is not None for c in cs):
                validity = torch.cat(
                    [
                        c.validity
                        if c.validity is not None
                        else torch.ones(len(c), dtype=torch.uint8, device=bytes_.device)
                        for c in cs
                    ]
                )
            cols[f.name] = Column(f.dtype, offsets=offs, bytes_=bytes_, validity=validity)
    return Batch(schema, cols)
