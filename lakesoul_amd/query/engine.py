"""Tensor query engine: GROUP BY / aggregates / equi-joins / sorts over
``Batch`` columns — the execution layer under the SQL console.

MI355X-native replacement for the reference's DataFusion execution
(rust/lakesoul-datafusion/src/lakesoul_table/mod.rs:51): every operator
is expressed as batched tensor ops (factorize via torch.unique, segment
reductions via scatter_reduce/index_add, vectorized hash join via
sorted-code expansion), so the same code runs on HBM-resident columns on
GPU and on CPU for tests. Strings are handled natively through 8-byte
chunk keys (the GPU path uses the str_chunk_keys HIP kernel; merge_gpu.py
uses the same normalization for string PK sorts).

pandas appears nowhere here: the SQL layer converts only the FINAL
(small) result for display.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..io.batch import Batch, Column
from ..io.schema import Field, Schema


# ------------------------------------------------------------------ #
# string chunk keys
# ------------------------------------------------------------------ #

def _chunk_keys_cpu(offsets: np.ndarray, data: np.ndarray, chunk: int
                    ) -> np.ndarray:
    """int64 big-endian-order key of bytes [8*chunk, 8*chunk+8) per row
    (shorter rows zero-padded) — order-preserving within the chunk."""
    n = len(offsets) - 1
    starts = offsets[:-1].astype(np.int64) + 8 * chunk
    lens = offsets[1:].astype(np.int64) - offsets[:-1].astype(np.int64)
    out = np.zeros(n, dtype=np.uint64)
    avail = np.clip(lens - 8 * chunk, 0, 8)
    for j in range(8):
        have = avail > j
        idx = np.where(have, starts + j, 0)
        b = np.where(have, data[np.clip(idx, 0, len(data) - 1 if len(data) else 0)], 0)
        out |= b.astype(np.uint64) << np.uint64(8 * (7 - j))
    return out.view(np.int64)


_SIGN64 = -0x8000000000000000


def _str_chunk_keys(col: Column, chunk: int) -> torch.Tensor:
    """RAW big-endian chunk keys (unsigned order = XOR sign bit before a
    signed argsort). The HIP kernel returns keys pre-flipped for direct
    signed sorts — undo that here so both paths agree."""
    offs = col.offsets
    if offs.device.type == "cuda":
        from ..ops import hip

        k = hip().str_chunk_keys(offs.to(torch.int64), col.bytes_, chunk)
        return k ^ torch.tensor(_SIGN64, dtype=torch.int64, device=k.device)
    o = offs.cpu().numpy().astype(np.int64)
    b = col.bytes_.cpu().numpy()
    return torch.from_numpy(_chunk_keys_cpu(o, b, chunk))


def _max_len(col: Column) -> int:
    offs = col.offsets
    if len(offs) <= 1:
        return 0
    lens = offs[1:] - offs[:-1]
    return int(lens.max().item())


# ------------------------------------------------------------------ #
# factorize
# ------------------------------------------------------------------ #

def _as_sortable_i64(keys: torch.Tensor) -> torch.Tensor:
    return keys


def _factorize_one(col: Column) -> Tuple[torch.Tensor, int]:
    """codes (n,) int64 in [0, g); group codes are ORDER-CONSISTENT with
    SQL ordering (nulls first in code space, flagged separately)."""
    n = len(col)
    if col.is_string:
        dev = col.offsets.device
        codes = torch.zeros(n, dtype=torch.int64, device=dev)
        ml = _max_len(col)
        nchunks = max(1, (ml + 7) // 8)
        for c in range(nchunks):
            k = _str_chunk_keys(col, c).to(dev)
            # order-consistent combine: sort by (codes, k)
            _, inv = torch.unique(k, sorted=True, return_inverse=True)
            ng = int(inv.max().item()) + 1 if n else 1
            codes = codes * ng + inv
            _, codes = torch.unique(codes, sorted=True, return_inverse=True)
    else:
        t = col.data
        if t.dtype == torch.uint8:
            t = t.to(torch.int64)
        _, codes = torch.unique(t, sorted=True, return_inverse=True)
        codes = codes.to(torch.int64)
    # nulls become their own (lowest) group
    if col.validity is not None:
        valid = col.validity.to(torch.bool).to(codes.device)
        codes = torch.where(valid, codes + 1, torch.zeros_like(codes))
        _, codes = torch.unique(codes, sorted=True, return_inverse=True)
    g = int(codes.max().item()) + 1 if n else 0
    return codes, g


def factorize(cols: Sequence[Column]) -> Tuple[torch.Tensor, int, torch.Tensor]:
    """Combined group codes over multiple columns.

    Returns (codes (n,) int64, n_groups, rep_idx (g,) int64 = first
    original row of each group). Codes are lexicographically
    order-consistent with the column tuple ordering (nulls first)."""
    if not cols:
        raise ValueError("factorize needs at least one column")
    n = len(cols[0])
    codes = None
    for c in cols:
        ci, gi = _factorize_one(c)
        if codes is None:
            codes = ci
        else:
            codes = codes * max(gi, 1) + ci.to(codes.device)
            _, codes = torch.unique(codes, sorted=True, return_inverse=True)
    g = int(codes.max().item()) + 1 if n else 0
    if codes.device.type == "cuda":
        # first-row-per-group WITHOUT scatter_reduce: amin into g≈few
        # slots serializes on atomics (measured 57 ms for 6M rows → 6
        # groups on MI355X; stable-argsort boundaries run in 0.7 ms)
        order = torch.argsort(codes, stable=True)
        cs = codes[order]
        first = torch.ones_like(cs, dtype=torch.bool)
        if n > 1:
            first[1:] = cs[1:] != cs[:-1]
        rep = order[first]
    else:
        rep = torch.full((g,), n, dtype=torch.int64, device=codes.device)
        rep.scatter_reduce_(0, codes, torch.arange(n, dtype=torch.int64,
                                                   device=codes.device),
                            reduce="amin", include_self=True)
    return codes, g, rep


# ------------------------------------------------------------------ #
# aggregates
# ------------------------------------------------------------------ #

_FLOAT_DTYPES = ("float32", "float64")


def _valid_mask(col: Column, n: int) -> torch.Tensor:
    if col.validity is None:
        dev = (col.offsets if col.is_string else col.data).device
        return torch.ones(n, dtype=torch.bool, device=dev)
    return col.validity.to(torch.bool)


def groupby_agg(batch: Batch, group_cols: Sequence[str],
                aggs: Sequence[Tuple[str, Optional[str], str, bool]]) -> Batch:
    """Aggregate ``batch`` by ``group_cols``.

    aggs: (fn, col_or_None, out_name, distinct) with fn in
    count/sum/min/max/avg. Output batch columns: group cols first (their
    representative values), then aggregate outputs. Groups come out in
    ascending group-key order (SQL engines don't guarantee order; ours is
    deterministic)."""
    from ..utils import timing as _tm

    n = batch.num_rows
    if group_cols:
        with _tm.phase("gb_factorize", sync_gpu=True):
            codes, g, rep = factorize([batch.columns[c] for c in group_cols])
    else:
        dev = None
        for c in batch.columns.values():
            t = c.offsets if c.offsets is not None else c.data
            if t is None:  # struct/map columns carry tensors in children
                continue
            dev = t.device
            break
        codes = torch.zeros(n, dtype=torch.int64, device=dev or "cpu")
        g, rep = 1, torch.zeros(1, dtype=torch.int64, device=dev or "cpu")

    out_fields: List[Field] = []
    out_cols: Dict[str, Column] = {}
    for c in group_cols:
        f = batch.schema.field(c)
        out_fields.append(Field(f.name, f.dtype, f.nullable))
        out_cols[c] = batch.columns[c].take(rep)

    # few-group aggregations on GPU: scatter_reduce into a handful of
    # slots serializes on atomics (every element CASes the same address);
    # per-group masked reductions are g clean passes instead
    small_g = (g <= 128 and g > 0 and codes.device.type == "cuda")
    with _tm.phase("gb_masks", sync_gpu=True):
        group_masks = (codes == torch.arange(g, device=codes.device)[:, None]
                       ) if small_g else None  # (g, n) bool

    def seg_count(valid_mask):
        if small_g:
            return (group_masks & valid_mask[None, :]).sum(1)
        return torch.bincount(codes[valid_mask], minlength=g)

    def seg_sum(v64, valid_mask, zero, skip_where=False):
        sel = v64 if skip_where else torch.where(valid_mask, v64, zero)
        if small_g:
            return torch.stack([
                torch.where(group_masks[j], sel, zero).sum()
                for j in range(g)])
        s = torch.zeros(g, dtype=v64.dtype, device=codes.device)
        s.index_add_(0, codes, sel)
        return s

    def seg_minmax(v64, valid_mask, sent, is_min):
        sel = torch.where(valid_mask, v64, torch.full_like(v64, sent))
        if small_g:
            big = torch.where(group_masks, sel[None, :],
                              torch.full((1, 1), sent, dtype=v64.dtype,
                                         device=v64.device))
            return big.amin(1) if is_min else big.amax(1)
        red = torch.full((g,), sent, dtype=v64.dtype, device=codes.device)
        red.scatter_reduce_(0, codes, sel, reduce="amin" if is_min else "amax",
                            include_self=True)
        return red

    counts_all = (group_masks.sum(1) if small_g
                  else torch.bincount(codes, minlength=g))

    # shared work across aggregates: a TPC-H q1-style SELECT asks for
    # sum(x)+avg(x) on the same column and count()s on non-null columns —
    # segment sums and per-group counts are cached per source column so
    # each distinct reduction runs ONCE (this was a 4x q1_sql gap)
    _cnt_cache: Dict[str, torch.Tensor] = {}
    _sum_cache: Dict[str, torch.Tensor] = {}

    for agg_i, (fn, cname, out_name, distinct) in enumerate(aggs):
        if fn == "count" and cname is None:
            out_fields.append(Field(out_name, "int64", False))
            out_cols[out_name] = Column("int64", data=counts_all.to(torch.int64))
            continue
        col = batch.columns[cname]
        all_valid = col.validity is None
        valid = _valid_mask(col, n)
        if fn == "count":
            if distinct:
                vc, _ = _factorize_one(col)
                pair = codes[valid] * max(int(vc.max().item()) + 1 if n else 1, 1) + vc[valid]
                upair_codes = torch.unique(pair)
                # recover group of each unique pair
                gsz = max(int(vc.max().item()) + 1 if n else 1, 1)
                cnt = torch.bincount(upair_codes // gsz, minlength=g)
            else:
                cnt = counts_all if all_valid else _cnt_cache.get(cname)
                if cnt is None:
                    cnt = seg_count(valid)
                    _cnt_cache[cname] = cnt
            out_fields.append(Field(out_name, "int64", False))
            out_cols[out_name] = Column("int64", data=cnt.to(torch.int64))
            continue
        if col.is_string:
            if fn in ("min", "max"):
                # rank-based: global sort ranks, segment amin/amax
                order = sort_indices(Batch(Schema([batch.schema.field(cname)]),
                                           {cname: col}), [(cname, True)])
                ranks = torch.empty(n, dtype=torch.int64, device=order.device)
                ranks[order] = torch.arange(n, dtype=torch.int64,
                                            device=order.device)
                sent = n if fn == "min" else -1
                red = seg_minmax(ranks, valid, sent, fn == "min")
                has = (red != sent)
                src_rows = torch.zeros(g, dtype=torch.int64, device=codes.device)
                src_rows[has] = order[red[has]]  # rank r -> row order[r]
                res = col.take(src_rows)
                if not bool(has.all()):
                    res.validity = has.to(torch.uint8)
                out_fields.append(Field(out_name, col.dtype, True))
                out_cols[out_name] = res
                continue
            raise ValueError(f"aggregate {fn} unsupported on string column")
        vals = col.data
        is_float = batch.schema.field(cname).dtype in _FLOAT_DTYPES
        acc_dtype = torch.float64 if is_float else torch.int64
        v64 = vals.to(acc_dtype)
        zero = torch.zeros_like(v64)
        if all_valid:
            vcnt = counts_all
        else:
            vcnt = _cnt_cache.get(cname)
            if vcnt is None:
                vcnt = seg_count(valid)
                _cnt_cache[cname] = vcnt
        if fn in ("sum", "avg"):
            s = _sum_cache.get(cname)
            if s is None:
                with _tm.phase("gb_seg_sum", sync_gpu=True):
                    s = seg_sum(v64, valid, zero, skip_where=all_valid)
                _sum_cache[cname] = s
            if fn == "avg":
                dt = "float64"
                res_t = (s.to(torch.float64)
                         / vcnt.clamp_min(1).to(torch.float64))
                fdt = batch.schema.field(cname).dtype
                if fdt.startswith("decimal"):
                    from ..io.schema import decimal_params

                    _, sc = decimal_params(fdt)
                    res_t = res_t / (10 ** sc)
                out_fields.append(Field(out_name, dt, True))
                out_cols[out_name] = Column(
                    dt, data=res_t,
                    validity=((vcnt > 0).to(torch.uint8)
                              if bool((vcnt == 0).any()) else None))
            else:
                fdt = batch.schema.field(cname).dtype
                dt = (fdt if fdt.startswith("decimal")
                      else ("float64" if is_float else "int64"))
                out_fields.append(Field(out_name, dt, True))
                out_cols[out_name] = Column(
                    dt, data=(s if not is_float else s.to(torch.float64)),
                    validity=((vcnt > 0).to(torch.uint8)
                              if bool((vcnt == 0).any()) else None))
        elif fn in ("min", "max"):
            if is_float:
                sent = float("inf") if fn == "min" else -float("inf")
            else:
                sent = (torch.iinfo(torch.int64).max if fn == "min"
                        else torch.iinfo(torch.int64).min)
            red = seg_minmax(v64, valid, sent, fn == "min")
            fdt = batch.schema.field(cname).dtype
            tdt = torch_dtype_of(fdt)
            out_fields.append(Field(out_name, fdt, True))
            out_cols[out_name] = Column(
                fdt, data=red.to(tdt),
                validity=((vcnt > 0).to(torch.uint8)
                          if bool((vcnt == 0).any()) else None))
        else:
            raise ValueError(f"unknown aggregate {fn}")
    return Batch(Schema(out_fields), out_cols)


def torch_dtype_of(dtype: str):
    from ..io.batch import torch_dtype_for

    return torch_dtype_for(dtype)


# ------------------------------------------------------------------ #
# join
# ------------------------------------------------------------------ #

def _concat_cols(a: Column, b: Column) -> Column:
    """Concatenate two columns of the same dtype (for joint factorize)."""
    na, nb = len(a), len(b)
    val = None
    if a.validity is not None or b.validity is not None:
        va = a.validity if a.validity is not None else torch.ones(na, dtype=torch.uint8)
        vb = b.validity if b.validity is not None else torch.ones(nb, dtype=torch.uint8)
        val = torch.cat([va, vb.to(va.device)])
    if a.is_string:
        offs = torch.cat([
            a.offsets.to(torch.int64),
            (b.offsets[1:].to(torch.int64) + int(a.offsets[-1])).to(a.offsets.device)])
        bys = torch.cat([a.bytes_, b.bytes_.to(a.bytes_.device)])
        return Column(a.dtype, offsets=offs, bytes_=bys, validity=val)
    return Column(a.dtype, data=torch.cat([a.data, b.data.to(a.data.device)]),
                  validity=val)


def hash_join(left: Batch, right: Batch, left_on: Sequence[str],
              right_on: Sequence[str], how: str = "inner"
              ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Vectorized equi-join. Returns (l_idx, r_idx) row index tensors;
    for how='left' unmatched left rows appear with r_idx == -1. NULL join
    keys never match (SQL semantics)."""
    nl, nr = left.num_rows, right.num_rows
    joint = [_concat_cols(left.columns[lc], right.columns[rc])
             for lc, rc in zip(left_on, right_on)]
    codes, g, _ = factorize(joint)
    lc, rc = codes[:nl], codes[nl:]
    # null keys never match
    def _null_any(b: Batch, names) -> torch.Tensor:
        m = None
        for c in names:
            col = b.columns[c]
            if col.validity is not None:
                nv = ~col.validity.to(torch.bool)
                m = nv if m is None else (m | nv)
        return m

    lnull = _null_any(left, left_on)
    rnull = _null_any(right, right_on)
    if rnull is not None:
        rc = torch.where(rnull.to(rc.device), torch.full_like(rc, -1), rc)
    valid_l = torch.ones(nl, dtype=torch.bool, device=lc.device)
    if lnull is not None:
        valid_l &= ~lnull.to(lc.device)

    order_r = torch.argsort(rc, stable=True)
    rs = rc[order_r]
    cnt = torch.bincount(rc[rc >= 0], minlength=g) if nr else torch.zeros(
        g, dtype=torch.int64, device=rc.device)
    off = torch.zeros(g + 1, dtype=torch.int64, device=rc.device)
    torch.cumsum(cnt, 0, out=off[1:].view(-1))
    n_neg = int((rs < 0).sum().item()) if nr else 0  # excluded rows sort first
    cnt_l = torch.where(valid_l, cnt[lc.clamp_min(0)], torch.zeros_like(lc))
    total = int(cnt_l.sum().item())
    out_l = torch.repeat_interleave(
        torch.arange(nl, dtype=torch.int64, device=lc.device), cnt_l)
    csum = torch.zeros(nl + 1, dtype=torch.int64, device=lc.device)
    torch.cumsum(cnt_l, 0, out=csum[1:].view(-1))
    pos = torch.arange(total, dtype=torch.int64, device=lc.device) - csum[out_l]
    out_r = order_r[n_neg + off[lc.clamp_min(0)][out_l] + pos]
    if how == "left":
        unmatched = (cnt_l == 0)
        extra_l = torch.arange(nl, dtype=torch.int64, device=lc.device)[unmatched]
        out_l = torch.cat([out_l, extra_l])
        out_r = torch.cat([out_r, torch.full((len(extra_l),), -1,
                                             dtype=torch.int64, device=lc.device)])
        order = torch.argsort(out_l, stable=True)
        out_l, out_r = out_l[order], out_r[order]
    elif how != "inner":
        raise ValueError(f"unsupported join kind {how}")
    return out_l, out_r


def join_batches(left: Batch, right: Batch, left_on, right_on, how,
                 rename_right: Optional[Dict[str, str]] = None) -> Batch:
    """Materialize the join result: left columns + right columns (right
    optionally renamed); right side of unmatched left rows is null."""
    l_idx, r_idx = hash_join(left, right, left_on, right_on, how)
    rename_right = rename_right or {}
    fields: List[Field] = []
    cols: Dict[str, Column] = {}
    for f in left.schema:
        fields.append(Field(f.name, f.dtype, f.nullable))
        cols[f.name] = left.columns[f.name].take(l_idx)
    has_unmatched = bool((r_idx < 0).any()) if len(r_idx) else False
    safe_r = r_idx.clamp_min(0)
    for f in right.schema:
        out_name = rename_right.get(f.name, f.name)
        fields.append(Field(out_name, f.dtype, True if has_unmatched else f.nullable))
        c = right.columns[f.name].take(safe_r)
        if has_unmatched:
            miss = (r_idx < 0)
            v = (c.validity.to(torch.bool)
                 if c.validity is not None
                 else torch.ones(len(r_idx), dtype=torch.bool, device=miss.device))
            c.validity = (v & ~miss).to(torch.uint8)
        cols[out_name] = c
    return Batch(Schema(fields), cols)


# ------------------------------------------------------------------ #
# sort / distinct
# ------------------------------------------------------------------ #

def sort_indices(batch: Batch, by: Sequence[Tuple[str, bool]]) -> torch.Tensor:
    """Stable sort indices by [(col, ascending)] — nulls last. Strings
    sort bytewise via iterated 8-byte chunk keys (LSD)."""
    n = batch.num_rows
    dev = None
    for c in batch.columns.values():
        dev = (c.offsets if c.is_string else c.data).device
        break
    idx = torch.arange(n, dtype=torch.int64, device=dev)
    for name, asc in reversed(list(by)):
        col = batch.columns[name]
        if col.is_string:
            ml = _max_len(col)
            nchunks = max(1, (ml + 7) // 8)
            for c in range(nchunks - 1, -1, -1):
                keys = _str_chunk_keys(col, c).to(dev)
                # unsigned order: flip sign bit
                keys = keys ^ torch.tensor(-0x8000000000000000, dtype=torch.int64,
                                           device=dev)
                k = keys[idx]
                order = torch.argsort(k, stable=True, descending=not asc)
                idx = idx[order]
        else:
            k = col.data[idx]
            order = torch.argsort(k, stable=True, descending=not asc)
            idx = idx[order]
        if col.validity is not None:
            nulls = (~col.validity.to(torch.bool))[idx]
            order = torch.argsort(nulls.to(torch.int8), stable=True)
            idx = idx[order]
    return idx


def distinct_indices(batch: Batch, cols: Sequence[str]) -> torch.Tensor:
    """First-occurrence row index per distinct tuple, in first-seen order."""
    codes, g, rep = factorize([batch.columns[c] for c in cols])
    return torch.sort(rep).values
