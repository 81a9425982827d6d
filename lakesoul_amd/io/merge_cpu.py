"""CPU merge-on-read engine (numpy) — correctness reference.

Implements the reference's sorted-merge semantics
(``rust/lakesoul-io/src/physical_plan/merge/``): rows from K sorted delta
files are merged by primary key; rows with equal PK are reduced per column
by a merge operator (``merge_operator.rs:22-31``):

- UseLast           last row (newest file) wins
- UseLastNotNull    last non-null value
- SumAll            sum of all values in the group (null if any null)
- SumLast           sum of the last value per file (null if any null)
- JoinedAllBy*      delimiter-join of all values (null if any null)
- JoinedLastBy*     delimiter-join of last value per file

The GPU engine (csrc/hip/kernels.hip) implements the same contract; GPU
tests compare against this module.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional

import numpy as np


@dataclass
class NpColumn:
    dtype: str
    data: Optional[np.ndarray] = None
    offsets: Optional[np.ndarray] = None
    bytes_: Optional[np.ndarray] = None
    validity: Optional[np.ndarray] = None  # uint8, 1=valid

    @property
    def is_string(self) -> bool:
        # list<T> columns ride the byte-string machinery (offsets in
        # bytes over the raw element buffer; whole-value UseLast)
        return self.dtype in ("string", "binary") or self.dtype.startswith("list<")

    def __len__(self):
        return len(self.offsets) - 1 if self.is_string else len(self.data)

    def valid_mask(self, n) -> np.ndarray:
        if self.validity is None:
            return np.ones(n, dtype=bool)
        return self.validity.astype(bool)

    def string_list(self) -> list:
        b = self.bytes_.tobytes()
        o = self.offsets
        v = self.valid_mask(len(self))
        return [b[o[i]:o[i + 1]] if v[i] else None for i in range(len(self))]


def _concat_column(cols: List[NpColumn]) -> NpColumn:
    dtype = cols[0].dtype
    n_total = sum(len(c) for c in cols)
    if cols[0].is_string:
        bytes_ = np.concatenate([c.bytes_ for c in cols]) if n_total else np.empty(0, np.uint8)
        offs = np.zeros(n_total + 1, dtype=np.int64)
        pos, base = 0, 0
        for c in cols:
            n = len(c)
            offs[pos + 1 : pos + n + 1] = c.offsets[1:].astype(np.int64) + base
            base += int(c.offsets[-1])
            pos += n
        validity = None
        if any(c.validity is not None for c in cols):
            validity = np.concatenate(
                [c.validity if c.validity is not None else np.ones(len(c), np.uint8) for c in cols]
            )
        return NpColumn(dtype, None, offs, bytes_, validity)
    data = np.concatenate([c.data for c in cols])
    validity = None
    if any(c.validity is not None for c in cols):
        validity = np.concatenate(
            [c.validity if c.validity is not None else np.ones(len(c), np.uint8) for c in cols]
        )
    return NpColumn(dtype, data, None, None, validity)


def _sort_keys(pk_cols: List[NpColumn], n_total: int):
    """Return lexsort order by (pk..., original position)."""
    keys = [np.arange(n_total)]
    for c in reversed(pk_cols):
        if c.is_string:
            b = c.bytes_.tobytes()
            o = c.offsets
            keys.append(np.array([b[o[i]:o[i + 1]] for i in range(n_total)], dtype=object))
        else:
            keys.append(c.data)
    return np.lexsort(tuple(keys))


def merge_sorted_files(
    file_columns: List[Dict[str, NpColumn]],
    pk: List[str],
    merge_ops: Optional[Dict[str, str]] = None,
    cdc_column: Optional[str] = None,
    present: Optional[List[set]] = None,
) -> Dict[str, NpColumn]:
    """Merge K files' columns (each already sorted by PK) with dedup.

    ``file_columns`` is ordered oldest -> newest (snapshot order).
    ``present``: optional per-file set of column names physically present
    in the file. A file missing a column contributes no values for it
    (schema-evolution / partial-upsert semantics: the reference's merger
    only builds ranges from batches that contain the column, so UseLast
    picks the newest file that HAS the column — v2/record_batch_builder.rs
    ColumnMapping). Missing-column entries in ``file_columns`` must still
    be null-filled for alignment.
    """
    merge_ops = merge_ops or {}
    names = []
    for fc in file_columns:
        for k in fc.keys():
            if k not in names:
                names.append(k)
    nfiles = len(file_columns)
    counts = [len(next(iter(fc.values()))) for fc in file_columns]
    n_total = sum(counts)
    seq = np.concatenate(
        [np.full(c, i, dtype=np.int32) for i, c in enumerate(counts)]
    ) if n_total else np.empty(0, np.int32)

    cat = {name: _concat_column([fc[name] for fc in file_columns]) for name in names}
    pk_cols = [cat[p] for p in pk]
    order = _sort_keys(pk_cols, n_total)

    # group boundaries on sorted pk
    if n_total == 0:
        return {name: cat[name] for name in names}
    new_group = np.zeros(n_total, dtype=bool)
    new_group[0] = True
    for c in pk_cols:
        if c.is_string:
            vals = np.array(
                [x if x is not None else b"" for x in c.string_list()], dtype=object
            )[order]
            new_group[1:] |= vals[1:] != vals[:-1]
        else:
            v = c.data[order]
            new_group[1:] |= v[1:] != v[:-1]
    starts = np.flatnonzero(new_group)
    ends = np.append(starts[1:], n_total)
    ngroups = len(starts)
    last_idx = order[ends - 1]  # original index of last row per group

    seq_sorted = seq[order]
    # "last row per (group, file)" mask, for *Last-per-stream ops
    gb = np.cumsum(new_group) - 1
    last_of_stream = np.zeros(n_total, dtype=bool)
    last_of_stream[-1] = True
    last_of_stream[:-1] = (seq_sorted[1:] != seq_sorted[:-1]) | new_group[1:]

    out: Dict[str, NpColumn] = {}
    for name in names:
        col = cat[name]
        op = merge_ops.get(name, "UseLast")
        if name in pk:
            op = "UseLast"  # PK values are identical within group
        contrib_sorted = None
        if present is not None:
            pres = np.array([name in p for p in present], dtype=bool)
            if not pres.all():
                contrib_sorted = pres[seq_sorted]
        out[name] = _apply_op(
            col, op, order, starts, ends, gb, last_idx, last_of_stream, n_total,
            contrib_sorted,
        )

    if cdc_column and cdc_column in out:
        keep = _cdc_keep_mask(out[cdc_column])
        idx = np.flatnonzero(keep)
        out = {name: _take(out[name], idx) for name in names}
    return out


def _take(col: NpColumn, idx: np.ndarray) -> NpColumn:
    if col.is_string:
        o = col.offsets
        lens = (o[1:] - o[:-1])[idx]
        new_o = np.zeros(len(idx) + 1, dtype=np.int64)
        np.cumsum(lens, out=new_o[1:])
        b = col.bytes_
        new_b = np.empty(int(new_o[-1]), dtype=np.uint8)
        for i, si in enumerate(idx):
            new_b[new_o[i]:new_o[i + 1]] = b[o[si]:o[si + 1]]
        v = None if col.validity is None else col.validity[idx]
        return NpColumn(col.dtype, None, new_o, new_b, v)
    v = None if col.validity is None else col.validity[idx]
    return NpColumn(col.dtype, col.data[idx], None, None, v)


def _cdc_keep_mask(col: NpColumn) -> np.ndarray:
    vals = col.string_list()
    return np.array([v != b"delete" for v in vals], dtype=bool)


def _apply_op(col, op, order, starts, ends, gb, last_idx, last_of_stream, n_total,
              contrib_sorted=None):
    ngroups = len(starts)
    if op == "UseLast" and contrib_sorted is None:
        return _take(col, last_idx)

    valid_sorted = col.valid_mask(n_total)[order]

    if op == "UseLast":
        # partial-column: last row among files that HAVE the column
        idx_arr = np.where(contrib_sorted, np.arange(n_total), -1)
        last_c = np.maximum.reduceat(idx_arr, starts)
        has = last_c >= 0
        src = order[np.where(has, last_c, 0)]
        res = _take(col, src)
        if col.validity is not None or not has.all():
            res.validity = (res.valid_mask(ngroups) & has).astype(np.uint8)
        return res

    if op == "UseLastNotNull":
        eligible = valid_sorted if contrib_sorted is None else (valid_sorted & contrib_sorted)
        idx_arr = np.where(eligible, np.arange(n_total), -1)
        last_valid_sorted = np.maximum.reduceat(idx_arr, starts)
        has = last_valid_sorted >= 0
        src = order[np.where(has, last_valid_sorted, 0)]
        res = _take(col, src)
        validity = has.astype(np.uint8)
        if col.validity is not None or not has.all():
            res.validity = (res.valid_mask(ngroups) & has).astype(np.uint8)
        return res

    if op in ("SumAll", "SumLast"):
        data_sorted = col.data[order]
        if op == "SumAll":
            contrib = np.ones(n_total, dtype=bool)
        else:
            contrib = last_of_stream.copy()
        if contrib_sorted is not None:
            contrib &= contrib_sorted
        # null if any contributing value is null (reference macro behavior)
        any_null = np.zeros(ngroups, dtype=bool)
        bad = (~valid_sorted) & contrib
        if bad.any():
            any_null = np.bitwise_or.reduceat(bad, starts)
        vals = np.where(contrib & valid_sorted, data_sorted, 0)
        sums = np.add.reduceat(vals, starts)
        validity = (~any_null).astype(np.uint8)
        return NpColumn(
            col.dtype,
            sums.astype(col.data.dtype),
            None,
            None,
            validity if not validity.all() else None,
        )

    if op in ("JoinedAllByComma", "JoinedAllBySemicolon", "JoinedLastByComma", "JoinedLastBySemicolon"):
        delim = b"," if op.endswith("Comma") else b";"
        use_all = "All" in op
        strings = col.string_list()
        out_items: List[Optional[bytes]] = []
        for g in range(ngroups):
            parts = []
            is_null = False
            for i in range(starts[g], ends[g]):
                if contrib_sorted is not None and not contrib_sorted[i]:
                    continue
                if not use_all and not last_of_stream[i]:
                    continue
                v = strings[order[i]]
                if v is None:
                    is_null = True
                    break
                parts.append(v)
            out_items.append(None if is_null else delim.join(parts))
        offs = np.zeros(ngroups + 1, dtype=np.int64)
        offs[1:] = np.cumsum([len(x) if x else 0 for x in out_items])
        bys = np.frombuffer(b"".join(x for x in out_items if x), dtype=np.uint8).copy()
        validity = np.array([x is not None for x in out_items], dtype=np.uint8)
        return NpColumn(
            col.dtype, None, offs, bys, validity if not validity.all() else None
        )

    raise ValueError(f"unknown merge operator {op}")
