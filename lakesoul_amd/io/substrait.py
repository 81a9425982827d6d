"""Substrait filter intake: decode Substrait protobuf filter payloads into
the engine's filter IR (``io/filters.py`` Expr).

The reference accepts filters in three encodings (string DSL, Substrait
``Plan``, Substrait ``ExtendedExpression`` — ``rust/lakesoul-io/src/filter/
parser.rs:44-60``); Spark/Flink push filters as Substrait bytes through
the C ABI (``native-io/.../substrait/SubstraitUtil.java``). This module
implements the wire format directly (protobuf wire decoding is ~40 lines;
no generated stubs needed, which also keeps the C ABI surface free of a
protobuf dependency). Field numbers follow the substrait-io spec and were
cross-checked against bytes produced by pyarrow.substrait (Acero), see
tests/test_substrait.py.

Supported expression subset (what engines push as *filters*):
scalar functions and/or/not/equal/not_equal/gt/gte/lt/lte/is_null/
is_not_null, literals (bool/int/float/string/binary/date/timestamp/
decimal/null), field references, SingularOrList (IN), and casts
(unwrapped). Unsupported constructs raise SubstraitError so callers fall
back to scanning without pushdown rather than mispruning.
"""

from __future__ import annotations

import struct
from typing import Dict, List, Optional, Tuple

from .filters import And, Cmp, Expr, IsNull, Literal, Not, Or
from .schema import Schema


class SubstraitError(ValueError):
    pass


# ------------------------------------------------------------------ #
# protobuf wire format
# ------------------------------------------------------------------ #

def _rd_varint(b: bytes, i: int) -> Tuple[int, int]:
    v = s = 0
    while True:
        if i >= len(b):
            raise SubstraitError("truncated varint")
        x = b[i]
        i += 1
        v |= (x & 0x7F) << s
        if not x & 0x80:
            return v, i
        s += 7
        if s > 70:
            raise SubstraitError("varint too long")


def _fields(b: bytes):
    """Yield (field_number, wire_type, value) triples of one message.
    value: int for varint/fixed, bytes for length-delimited."""
    i = 0
    while i < len(b):
        tag, i = _rd_varint(b, i)
        fn, wt = tag >> 3, tag & 7
        if wt == 0:
            v, i = _rd_varint(b, i)
            yield fn, wt, v
        elif wt == 1:
            if i + 8 > len(b):
                raise SubstraitError("truncated fixed64")
            yield fn, wt, int.from_bytes(b[i:i + 8], "little")
            i += 8
        elif wt == 2:
            ln, i = _rd_varint(b, i)
            if i + ln > len(b):
                raise SubstraitError("truncated bytes")
            yield fn, wt, b[i:i + ln]
            i += ln
        elif wt == 5:
            if i + 4 > len(b):
                raise SubstraitError("truncated fixed32")
            yield fn, wt, int.from_bytes(b[i:i + 4], "little")
            i += 4
        else:
            raise SubstraitError(f"unsupported wire type {wt}")


def _submsgs(b: bytes, field: int) -> List[bytes]:
    return [v for fn, wt, v in _fields(b) if fn == field and wt == 2]


def _first(b: bytes, field: int) -> Optional[bytes]:
    for fn, wt, v in _fields(b):
        if fn == field and wt == 2:
            return v
    return None


def _varint_field(b: bytes, field: int) -> Optional[int]:
    for fn, wt, v in _fields(b):
        if fn == field and wt == 0:
            return v
    return None


def _zigzag(v: int) -> int:
    return (v >> 1) ^ -(v & 1)


def _signed(v: int, bits: int) -> int:
    if v >= 1 << (bits - 1):
        v -= 1 << bits
    return v


# ------------------------------------------------------------------ #
# substrait message interpretation
# ------------------------------------------------------------------ #

# simple-extension function names -> engine ops
_CMP_FUNCS = {
    "equal": "eq",
    "not_equal": "noteq",
    "gt": "gt",
    "gte": "gteq",
    "lt": "lt",
    "lte": "lteq",
}
_CMP_SWAP = {"eq": "eq", "noteq": "noteq", "gt": "lt", "gteq": "lteq",
             "lt": "gt", "lteq": "gteq"}


def _function_names(root: bytes) -> Dict[int, str]:
    """anchor -> simple name from SimpleExtensionDeclaration list (field 2
    of Plan / ExtendedExpression)."""
    out: Dict[int, str] = {}
    for decl in _submsgs(root, 2):
        ext_fn = _first(decl, 3)  # extension_function
        if ext_fn is None:
            continue
        anchor = _varint_field(ext_fn, 2) or 0
        name_b = _first(ext_fn, 3)
        name = name_b.decode() if name_b is not None else ""
        # composite names are "name:sig"
        out[anchor] = name.split(":", 1)[0]
    return out


def _schema_names(named_struct: Optional[bytes], fallback: Schema) -> List[str]:
    if named_struct is not None:
        names = [v.decode() for fn, wt, v in _fields(named_struct)
                 if fn == 1 and wt == 2]
        if names:
            return names
    return fallback.names()


class _Ctx:
    def __init__(self, funcs: Dict[int, str], names: List[str], schema: Schema):
        self.funcs = funcs
        self.names = names
        self.schema = schema


def _literal_value(lit: bytes):
    """Expression.Literal -> python value (None for the null literal)."""
    for fn, wt, v in _fields(lit):
        if fn == 1 and wt == 0:      # boolean
            return bool(v)
        if fn == 2 and wt == 0:      # i8
            return _signed(v, 64)
        if fn == 3 and wt == 0:      # i16
            return _signed(v, 64)
        if fn == 5 and wt == 0:      # i32
            return _signed(v, 64)
        if fn == 7 and wt == 0:      # i64
            return _signed(v, 64)
        if fn == 10 and wt == 5:     # fp32
            return struct.unpack("<f", v.to_bytes(4, "little"))[0]
        if fn == 11 and wt == 1:     # fp64
            return struct.unpack("<d", v.to_bytes(8, "little"))[0]
        if fn == 12 and wt == 2:     # string
            return v.decode()
        if fn == 13 and wt == 2:     # binary
            return bytes(v)
        if fn == 14 and wt == 0:     # timestamp (us)
            return _signed(v, 64)
        if fn == 16 and wt == 0:     # date (days)
            return _signed(v, 64)
        if fn == 17 and wt == 0:     # time (us)
            return _signed(v, 64)
        if fn == 22 and wt == 2:     # var_char {value=1, length=2}
            s = _first(v, 1)
            return s.decode() if s is not None else ""
        if fn == 21 and wt == 2:     # fixed_char
            return v.decode()
        if fn == 24 and wt == 2:     # decimal {value(16B LE)=1, precision=2, scale=3}
            raw = _first(v, 1) or b""
            scale = _varint_field(v, 3) or 0
            unscaled = int.from_bytes(raw, "little", signed=True)
            return unscaled / (10 ** scale) if scale else unscaled
        if fn == 27 and wt == 2:     # timestamp_tz — not emitted as submsg
            return v
        if fn == 29 and wt == 2:     # null (typed)
            return None
    raise SubstraitError("unsupported literal")


def _field_index(sel: bytes) -> int:
    """FieldReference -> struct field index (direct_reference chain)."""
    seg = _first(sel, 1)  # direct_reference: ReferenceSegment
    if seg is None:
        raise SubstraitError("unsupported field reference (no direct ref)")
    sf = _first(seg, 2)   # struct_field
    if sf is not None:
        return _varint_field(sf, 1) or 0
    mk = _first(seg, 1)   # map_key {map_key: Literal}
    if mk is not None:
        raise SubstraitError("map-key field reference carries a name, "
                             "resolve via _field_name")
    raise SubstraitError("unsupported reference segment")


def _field_name(sel: bytes, ctx: _Ctx) -> str:
    seg = _first(sel, 1)
    if seg is not None:
        mk = _first(seg, 1)
        if mk is not None:
            lit = _first(mk, 1)
            if lit is not None:
                v = _literal_value(lit)
                if isinstance(v, str):
                    return v
    idx = _field_index(sel)
    if idx >= len(ctx.names):
        raise SubstraitError(f"field index {idx} out of range")
    return ctx.names[idx]


def _expr(e: bytes, ctx: _Ctx) -> Expr:
    """Expression message -> engine Expr (boolean-valued)."""
    lit = _first(e, 1)
    if lit is not None:
        v = _literal_value(lit)
        if isinstance(v, bool):
            return Literal(v)
        raise SubstraitError("non-boolean literal at predicate position")
    fn_msg = _first(e, 3)
    if fn_msg is not None:
        return _scalar_function(fn_msg, ctx)
    sol = _first(e, 8)   # SingularOrList
    if sol is not None:
        return _singular_or_list(sol, ctx)
    cast = _first(e, 11)
    if cast is not None:
        inner = _first(cast, 2)
        if inner is None:
            raise SubstraitError("cast without input")
        return _expr(inner, ctx)
    raise SubstraitError("unsupported expression")


def _value_operand(e: bytes, ctx: _Ctx):
    """Expression at argument position -> ('col', name) | ('lit', value)."""
    sel = _first(e, 2)
    if sel is not None:
        return ("col", _field_name(sel, ctx))
    lit = _first(e, 1)
    if lit is not None:
        return ("lit", _literal_value(lit))
    cast = _first(e, 11)
    if cast is not None:
        inner = _first(cast, 2)
        if inner is not None:
            return _value_operand(inner, ctx)
    raise SubstraitError("unsupported operand")


def _fn_args(fn_msg: bytes) -> List[bytes]:
    """ScalarFunction arguments: FunctionArgument.value Expressions
    (field 4), plus deprecated direct args (field 2)."""
    out = []
    for arg in _submsgs(fn_msg, 4):
        v = _first(arg, 3)
        if v is not None:
            out.append(v)
    out.extend(_submsgs(fn_msg, 2))  # pre-0.9 producers
    return out


def _scalar_function(fn_msg: bytes, ctx: _Ctx) -> Expr:
    anchor = _varint_field(fn_msg, 1) or 0
    name = ctx.funcs.get(anchor)
    if name is None:
        raise SubstraitError(f"unknown function anchor {anchor}")
    args = _fn_args(fn_msg)
    if name == "and" or name == "or":
        if len(args) < 2:
            raise SubstraitError(f"{name} needs >=2 args")
        sub = [_expr(a, ctx) for a in args]
        out = sub[0]
        for s in sub[1:]:
            out = And(out, s) if name == "and" else Or(out, s)
        return out
    if name == "not":
        if len(args) != 1:
            raise SubstraitError("not needs 1 arg")
        return Not(_expr(args[0], ctx))
    if name in ("is_null", "is_not_null"):
        if len(args) != 1:
            raise SubstraitError(f"{name} needs 1 arg")
        kind, v = _value_operand(args[0], ctx)
        if kind != "col":
            raise SubstraitError(f"{name} on non-column")
        return IsNull(v, negate=(name == "is_not_null"))
    if name in _CMP_FUNCS:
        if len(args) != 2:
            raise SubstraitError(f"{name} needs 2 args")
        (ka, va), (kb, vb) = _value_operand(args[0], ctx), _value_operand(args[1], ctx)
        op = _CMP_FUNCS[name]
        if ka == "col" and kb == "lit":
            col, val = va, vb
        elif ka == "lit" and kb == "col":
            col, val = vb, va
            op = _CMP_SWAP[op]
        else:
            raise SubstraitError(f"{name} must compare a column to a literal")
        if val is None:
            if op == "eq":
                return IsNull(col)
            if op == "noteq":
                return IsNull(col, negate=True)
            return Literal(True)
        return Cmp(col, op, _coerce(val, col, ctx))
    raise SubstraitError(f"unsupported function {name}")


def _singular_or_list(sol: bytes, ctx: _Ctx) -> Expr:
    value = _first(sol, 1)
    if value is None:
        raise SubstraitError("SingularOrList without value")
    kind, col = _value_operand(value, ctx)
    if kind != "col":
        raise SubstraitError("IN on non-column")
    opts = []
    for o in _submsgs(sol, 2):
        k, v = _value_operand(o, ctx)
        if k != "lit":
            raise SubstraitError("IN with non-literal option")
        opts.append(_coerce(v, col, ctx))
    return Cmp(col, "in", opts)


def _coerce(val, col: str, ctx: _Ctx):
    """Align the literal's python type with the column dtype (decimal
    literals stay logical; the evaluator scales them)."""
    try:
        f = ctx.schema.field(col)
    except KeyError:
        return val
    if f.dtype in ("float32", "float64") and isinstance(val, int):
        return float(val)
    if f.dtype.startswith("int") and isinstance(val, float) and val.is_integer():
        return int(val)
    if f.dtype in ("string",) and isinstance(val, bytes):
        return val.decode()
    if f.dtype == "binary" and isinstance(val, str):
        return val.encode()
    return val


# ------------------------------------------------------------------ #
# entry points
# ------------------------------------------------------------------ #

def _from_extended_expression(buf: bytes, schema: Schema) -> Expr:
    """ExtendedExpression: extension_uris=1 extensions=2 referred_expr=3
    base_schema=4."""
    refs = _submsgs(buf, 3)
    if not refs:
        raise SubstraitError("no referred_expr")
    names = _schema_names(_first(buf, 4), schema)
    ctx = _Ctx(_function_names(buf), names, schema)
    out: Optional[Expr] = None
    for ref in refs:
        e = _first(ref, 1)
        if e is None:
            raise SubstraitError("referred_expr without expression (measure?)")
        ex = _expr(e, ctx)
        out = ex if out is None else And(out, ex)
    return out


def _from_plan(buf: bytes, schema: Schema) -> Expr:
    """Plan: extensions=2 relations=3; relations[0] root(2)/rel(1) ->
    Rel.read(1) -> ReadRel{base_schema=2, filter=3}."""
    rels = _submsgs(buf, 3)
    if not rels:
        raise SubstraitError("plan has no relations")
    funcs = _function_names(buf)
    for plan_rel in rels:
        rel = _first(plan_rel, 2)  # root: RelRoot
        if rel is not None:
            rel = _first(rel, 1)   # input: Rel
        else:
            rel = _first(plan_rel, 1)
        if rel is None:
            continue
        read = _first(rel, 1)      # ReadRel
        if read is None:
            continue
        filt = _first(read, 3)
        if filt is None:
            filt = _first(read, 11)  # best_effort_filter
        if filt is None:
            continue
        names = _schema_names(_first(read, 2), schema)
        return _expr(filt, _Ctx(funcs, names, schema))
    raise SubstraitError("no ReadRel filter found in plan")


def decode_substrait_filter(buf: bytes, schema: Schema) -> Expr:
    """Parse Substrait bytes (ExtendedExpression or Plan — tried in that
    order like the reference, parser.rs:576-586) into a filter Expr."""
    if not isinstance(buf, (bytes, bytearray, memoryview)):
        raise TypeError("substrait filter must be bytes")
    buf = bytes(buf)
    errors = []
    try:
        return _from_extended_expression(buf, schema)
    except SubstraitError as e:
        errors.append(f"extended_expression: {e}")
    try:
        return _from_plan(buf, schema)
    except SubstraitError as e:
        errors.append(f"plan: {e}")
    raise SubstraitError("; ".join(errors))


# ------------------------------------------------------------------ #
# encoder (producer side: tests, C ABI round trip, connector tooling)
# ------------------------------------------------------------------ #

def _w_varint(v: int) -> bytes:
    out = bytearray()
    while True:
        b = v & 0x7F
        v >>= 7
        if v:
            out.append(b | 0x80)
        else:
            out.append(b)
            return bytes(out)


def _w_tag(fn: int, wt: int) -> bytes:
    return _w_varint((fn << 3) | wt)


def _w_len(fn: int, payload: bytes) -> bytes:
    return _w_tag(fn, 2) + _w_varint(len(payload)) + payload


def _w_vint(fn: int, v: int) -> bytes:
    return _w_tag(fn, 0) + _w_varint(v)


_TYPE_FIELD = {
    "bool": 1, "int8": 2, "int16": 3, "int32": 5, "int64": 7,
    "float32": 10, "float64": 11, "string": 12, "binary": 13,
    "timestamp": 14, "date": 16,
}


def _enc_literal(v) -> bytes:
    if isinstance(v, bool):
        return _w_vint(1, int(v))
    if isinstance(v, int):
        return _w_vint(7, v & ((1 << 64) - 1))
    if isinstance(v, float):
        return _w_tag(11, 1) + struct.pack("<d", v)
    if isinstance(v, str):
        return _w_len(12, v.encode())
    if isinstance(v, (bytes, bytearray)):
        return _w_len(13, bytes(v))
    if v is None:
        # null literal: typed null (i64, nullable)
        return _w_len(29, _w_len(7, _w_vint(2, 1)))
    raise SubstraitError(f"cannot encode literal {type(v)}")


def _enc_field_ref(idx: int) -> bytes:
    sf = _w_vint(1, idx) if idx else b""
    seg = _w_len(2, sf)
    return _w_len(1, seg) + _w_len(4, b"")  # direct_reference + root_reference


def _enc_expr_col(idx: int) -> bytes:
    return _w_len(2, _enc_field_ref(idx))  # Expression.selection


def _enc_expr_lit(v) -> bytes:
    return _w_len(1, _enc_literal(v))      # Expression.literal


def _enc_fn(anchor: int, arg_exprs: List[bytes]) -> bytes:
    body = _w_vint(1, anchor)
    for a in arg_exprs:
        body += _w_len(4, _w_len(3, a))    # FunctionArgument.value
    return _w_len(3, body)                 # Expression.scalar_function


class _FnTable:
    def __init__(self):
        self.anchors: Dict[str, int] = {}

    def anchor(self, name: str) -> int:
        if name not in self.anchors:
            self.anchors[name] = len(self.anchors)
        return self.anchors[name]


_INV_CMP = {v: k for k, v in _CMP_FUNCS.items()}


def _enc_pred(e: Expr, schema: Schema, ft: _FnTable) -> bytes:
    names = schema.names()
    if isinstance(e, And) or isinstance(e, Or):
        nm = "and" if isinstance(e, And) else "or"
        return _enc_fn(ft.anchor(nm),
                       [_enc_pred(e.left, schema, ft), _enc_pred(e.right, schema, ft)])
    if isinstance(e, Not):
        return _enc_fn(ft.anchor("not"), [_enc_pred(e.inner, schema, ft)])
    if isinstance(e, IsNull):
        nm = "is_not_null" if e.negate else "is_null"
        return _enc_fn(ft.anchor(nm), [_enc_expr_col(names.index(e.col))])
    if isinstance(e, Literal):
        return _enc_expr_lit(bool(e.value))
    if isinstance(e, Cmp):
        idx = names.index(e.col)
        if e.op == "in":
            body = _w_len(1, _enc_expr_col(idx))
            for v in e.value:
                body += _w_len(2, _enc_expr_lit(v))
            return _w_len(8, body)         # Expression.singular_or_list
        return _enc_fn(ft.anchor(_INV_CMP[e.op]),
                       [_enc_expr_col(idx), _enc_expr_lit(e.value)])
    raise SubstraitError(f"cannot encode {type(e).__name__}")


def _enc_named_struct(schema: Schema) -> bytes:
    body = b""
    for n in schema.names():
        body += _w_len(1, n.encode())
    types = b""
    for f in schema:
        tf = _TYPE_FIELD.get(f.dtype, 12)
        types += _w_len(1, _w_len(tf, _w_vint(2, 1 if f.nullable else 2)))
    body += _w_len(2, types)  # Type.Struct{types=1}
    return body


def _enc_extensions(ft: _FnTable) -> bytes:
    out = _w_len(1, _w_vint(1, 1) + _w_len(2, b"urn:lakesoul:functions"))
    for name, anchor in ft.anchors.items():
        ext_fn = _w_vint(1, 1) + _w_vint(2, anchor) + _w_len(3, name.encode())
        out += _w_len(2, _w_len(3, ext_fn))
    return out


def encode_substrait_filter(e: Expr, schema: Schema) -> bytes:
    """Encode a filter Expr as a Substrait ExtendedExpression (the same
    shape pyarrow/Acero and the reference's Java SubstraitUtil produce)."""
    ft = _FnTable()
    pred = _enc_pred(e, schema, ft)
    ref = _w_len(1, pred) + _w_len(3, b"filter")
    return (_enc_extensions(ft)
            + _w_len(3, ref)
            + _w_len(4, _enc_named_struct(schema)))


def encode_substrait_plan_filter(e: Expr, schema: Schema) -> bytes:
    """Encode a filter Expr as a Substrait Plan whose single relation is a
    ReadRel carrying the filter (what Spark pushes, parser.rs:513-522)."""
    ft = _FnTable()
    pred = _enc_pred(e, schema, ft)
    read_rel = (_w_len(2, _enc_named_struct(schema))   # base_schema
                + _w_len(3, pred)                      # filter
                + _w_len(7, _w_len(1, b"t")))          # named_table{names}
    rel = _w_len(1, read_rel)                          # Rel.read
    root = _w_len(1, rel)                              # RelRoot.input
    plan_rel = _w_len(2, root)                         # PlanRel.root
    return _enc_extensions(ft) + _w_len(3, plan_rel)
