"""Filter expressions: legacy string-DSL parsing, evaluation, pruning.

Implements the reference's Java filter string DSL
(``rust/lakesoul-io/src/filter/parser.rs:52-120``):
``and(l,r) or(l,r) not(e) eq(col,v) noteq(col,v) gt/gteq/lt/lteq(col,v)``
with ``null`` literals for is-null tests; plus python tuple filters
``(col, op, value)``.

The same expression drives three stages, mirroring the reference's
filter classification (session.rs:760-791):
1. partition pruning (range-partition columns vs partition_desc),
2. file/row-group pruning from parquet min/max statistics,
3. exact row-level evaluation on decoded batches (torch masks, GPU or CPU).
"""

from __future__ import annotations

import struct
from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple, Union

import numpy as np
import torch

from .schema import Schema


class Expr:
    def evaluate(self, batch) -> torch.Tensor:  # bool mask
        raise NotImplementedError

    def prune_stats(self, stats: Dict[str, Tuple]) -> bool:
        """False = no row can match (skip the file/row group)."""
        return True

    def partition_prune(self, part_values: Dict[str, str]) -> bool:
        """False = partition cannot match."""
        return True

    def pk_eq_values(self) -> Dict[str, object]:
        """col -> constant for top-level AND-ed equality predicates."""
        return {}

    def columns(self) -> set:
        """All column names referenced by this expression."""
        out = set()
        for f in getattr(self, "__dataclass_fields__", {}):
            v = getattr(self, f)
            if isinstance(v, Expr):
                out |= v.columns()
        if hasattr(self, "col"):
            out.add(self.col)
        return out


@dataclass
class Literal(Expr):
    value: bool

    def evaluate(self, batch):
        n = batch.num_rows
        dev = _batch_device(batch)
        return torch.full((n,), self.value, dtype=torch.bool, device=dev)

    def prune_stats(self, stats):
        return self.value

    def partition_prune(self, pv):
        return self.value


@dataclass
class And(Expr):
    left: Expr
    right: Expr

    def evaluate(self, batch):
        return self.left.evaluate(batch) & self.right.evaluate(batch)

    def prune_stats(self, stats):
        return self.left.prune_stats(stats) and self.right.prune_stats(stats)

    def partition_prune(self, pv):
        return self.left.partition_prune(pv) and self.right.partition_prune(pv)

    def pk_eq_values(self):
        d = dict(self.left.pk_eq_values())
        d.update(self.right.pk_eq_values())
        return d


@dataclass
class Or(Expr):
    left: Expr
    right: Expr

    def evaluate(self, batch):
        return self.left.evaluate(batch) | self.right.evaluate(batch)

    def prune_stats(self, stats):
        return self.left.prune_stats(stats) or self.right.prune_stats(stats)

    def partition_prune(self, pv):
        return self.left.partition_prune(pv) or self.right.partition_prune(pv)


@dataclass
class Not(Expr):
    inner: Expr

    def evaluate(self, batch):
        return ~self.inner.evaluate(batch)


@dataclass
class IsNull(Expr):
    col: str
    negate: bool = False

    def evaluate(self, batch):
        c = batch.columns[self.col]
        n = batch.num_rows
        dev = _batch_device(batch)
        if c.validity is None:
            m = torch.zeros(n, dtype=torch.bool, device=dev)
        else:
            m = ~(c.validity.to(torch.bool))
        return ~m if self.negate else m

    def partition_prune(self, pv):
        if self.col not in pv:
            return True
        is_null = pv[self.col] is None
        return (not is_null) if self.negate else is_null


@dataclass
class Cmp(Expr):
    col: str
    op: str  # eq noteq gt gteq lt lteq in
    value: object

    def evaluate(self, batch):
        c = batch.columns[self.col]
        n = batch.num_rows
        if c.is_string:
            offs = c.offsets.cpu().numpy()
            bys = c.bytes_.cpu().numpy().tobytes()
            v = self.value.encode() if isinstance(self.value, str) else bytes(self.value)
            vals = [bys[offs[i]:offs[i + 1]] for i in range(n)]
            m = np.array([_cmp_py(x, self.op, v) for x in vals])
            out = torch.from_numpy(m).to(_batch_device(batch))
        else:
            t = c.data
            v = self.value
            if c.dtype.startswith("decimal") and not isinstance(v, (list, tuple, set)):
                # logical literal -> unscaled int64 (decimal columns store
                # unscaled values; exact scaling via decimal module)
                import decimal as _dec

                from .schema import decimal_params

                _, sc = decimal_params(c.dtype)
                v = int(_dec.Decimal(str(v)).scaleb(sc).to_integral_value())
            if self.op == "eq":
                out = t == v
            elif self.op == "noteq":
                out = t != v
            elif self.op == "gt":
                out = t > v
            elif self.op == "gteq":
                out = t >= v
            elif self.op == "lt":
                out = t < v
            elif self.op == "lteq":
                out = t <= v
            elif self.op == "in":
                out = torch.isin(t, torch.tensor(list(v), device=t.device))
            else:
                raise ValueError(self.op)
        if c.validity is not None:
            out = out & c.validity.to(torch.bool)
        return out

    def prune_stats(self, stats):
        if self.col not in stats:
            return True
        mn, mx = stats[self.col]
        if mn is None or mx is None:
            return True
        v = self.value
        try:
            if self.op == "eq":
                return mn <= v <= mx
            if self.op == "gt":
                return mx > v
            if self.op == "gteq":
                return mx >= v
            if self.op == "lt":
                return mn < v
            if self.op == "lteq":
                return mn <= v
            if self.op == "in":
                return any(mn <= x <= mx for x in v)
        except TypeError:
            return True
        return True

    def partition_prune(self, pv):
        if self.col not in pv:
            return True
        if pv[self.col] is None:
            # NULL partition value: comparisons are unknown -> no row of
            # this partition can satisfy the predicate
            return False
        try:
            col_v = type(self.value)(pv[self.col]) if not isinstance(self.value, str) else pv[self.col]
        except (TypeError, ValueError):
            return True
        return _cmp_py(col_v, self.op, self.value)

    def pk_eq_values(self):
        return {self.col: self.value} if self.op == "eq" else {}


def _cmp_py(a, op, b) -> bool:
    if op == "eq":
        return a == b
    if op == "noteq":
        return a != b
    if op == "gt":
        return a > b
    if op == "gteq":
        return a >= b
    if op == "lt":
        return a < b
    if op == "lteq":
        return a <= b
    if op == "in":
        return a in b
    raise ValueError(op)


def _batch_device(batch):
    for c in batch.columns.values():
        t = c.data if not c.is_string else c.bytes_
        if t is not None:
            return t.device
    return torch.device("cpu")


_TUPLE_OPS = {
    "==": "eq", "=": "eq", "!=": "noteq", ">": "gt", ">=": "gteq",
    "<": "lt", "<=": "lteq", "in": "in",
    "eq": "eq", "noteq": "noteq", "gt": "gt", "gteq": "gteq",
    "lt": "lt", "lteq": "lteq",
}


def from_tuples(filters: List[Tuple], schema: Schema) -> Optional[Expr]:
    expr: Optional[Expr] = None
    for col, op, val in filters:
        e: Expr = Cmp(col, _TUPLE_OPS[op], val)
        expr = e if expr is None else And(expr, e)
    return expr


def parse_filter_dsl(s: str, schema: Schema) -> Expr:
    """Parse the reference's Java filter string DSL (parser.rs:52-120)."""
    s = s.strip()
    i = s.find("(")
    if i < 0 or not s.endswith(")"):
        raise ValueError(f"bad filter string: {s}")
    op = s[:i].strip()
    body = s[i + 1 : -1]
    # split top-level comma
    k = 0
    split = -1
    for j, ch in enumerate(body):
        if ch == "(":
            k += 1
        elif ch == ")":
            k -= 1
        elif ch == "," and k == 0 and split < 0:
            split = j
    if op == "not":
        return Not(parse_filter_dsl(body, schema))
    if op in ("and", "or"):
        left, right = body[:split], body[split + 1 :]
        l, r = parse_filter_dsl(left, schema), parse_filter_dsl(right, schema)
        return And(l, r) if op == "and" else Or(l, r)
    if op in ("eq", "noteq", "gt", "gteq", "lt", "lteq"):
        col = body[:split].strip()
        rhs = body[split + 1 :].strip()
        if rhs == "null":
            if op == "eq":
                return IsNull(col)
            if op == "noteq":
                return IsNull(col, negate=True)
            return Literal(True)
        try:
            f = schema.field(col)
        except KeyError:
            return Literal(False)
        return Cmp(col, op, _parse_literal(rhs, f.dtype))
    raise ValueError(f"unknown filter op {op}")


def _parse_literal(s: str, dtype: str):
    if dtype.startswith("decimal"):
        return float(s)   # scaled at evaluate() against the unscaled column
    if dtype in ("string", "binary"):
        if len(s) >= 2 and s[0] == s[-1] and s[0] in "'\"":
            s = s[1:-1]
        return s
    if dtype in ("float32", "float64"):
        return float(s)
    if dtype == "bool":
        return s.lower() in ("true", "1")
    return int(s)


def resolve_filters(filters, schema: Schema) -> Optional[Expr]:
    """Accept None | Expr | DSL string | list of (col,op,val) tuples |
    Substrait protobuf bytes (Plan or ExtendedExpression — the encodings
    engines push through the C ABI, reference parser.rs:44-60)."""
    if filters is None:
        return None
    if isinstance(filters, Expr):
        return filters
    if isinstance(filters, (bytes, bytearray, memoryview)):
        from .substrait import decode_substrait_filter

        return decode_substrait_filter(bytes(filters), schema)
    if isinstance(filters, str):
        return parse_filter_dsl(filters, schema)
    if isinstance(filters, (list, tuple)):
        if not filters:
            return None
        return from_tuples(list(filters), schema)
    raise TypeError(f"unsupported filters: {filters!r}")


def decode_stat(raw: bytes, dtype: str):
    if raw is None:
        return None
    if dtype in ("int32", "int8", "int16", "date32"):
        return struct.unpack("<i", raw)[0]
    if dtype in ("int64", "timestamp[us]", "timestamp[ms]", "timestamp[ns]"):
        return struct.unpack("<q", raw)[0]
    if dtype == "float32":
        return struct.unpack("<f", raw)[0]
    if dtype == "float64":
        return struct.unpack("<d", raw)[0]
    if dtype == "string":
        return raw.decode("utf-8", "replace")
    if dtype == "binary":
        return raw
    return None
