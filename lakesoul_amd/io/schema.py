"""Table schema model + Spark-compatible JSON serde.

The reference stores ``table_info.table_schema`` as Spark-style schema JSON
(serialized from Arrow by ``rust/lakesoul-common/src/ser/arrow_java.rs``).
We emit/parse the same structure: ``{"type":"struct","fields":[{"name":...,
"type":<spark type name>,"nullable":...,"metadata":{}}]}``.
"""

from __future__ import annotations

import json
from dataclasses import dataclass, field as dc_field
from typing import List, Optional, Sequence, Tuple, Union

# canonical dtype strings used across the engine
_CANONICAL = {
    "bool": "bool",
    "boolean": "bool",
    "int8": "int8",
    "byte": "int8",
    "int16": "int16",
    "short": "int16",
    "int32": "int32",
    "int": "int32",
    "integer": "int32",
    "int64": "int64",
    "long": "int64",
    "float16": "float16",
    "half": "float16",
    "float32": "float32",
    "float": "float32",
    "float64": "float64",
    "double": "float64",
    "string": "string",
    "utf8": "string",
    "str": "string",
    "binary": "binary",
    "bytes": "binary",
    "date32": "date32",
    "date": "date32",
    "timestamp[us]": "timestamp[us]",
    "timestamp": "timestamp[us]",
    "timestamp[ms]": "timestamp[ms]",
    "timestamp[ns]": "timestamp[ns]",
}

_TO_SPARK = {
    "bool": "boolean",
    "int8": "byte",
    "int16": "short",
    "int32": "integer",
    "int64": "long",
    "float16": "half",
    "float32": "float",
    "float64": "double",
    "string": "string",
    "binary": "binary",
    "date32": "date",
    "timestamp[us]": "timestamp",
    "timestamp[ms]": "timestamp_millis",
    "timestamp[ns]": "timestamp_nanos",
}
_FROM_SPARK = {v: k for k, v in _TO_SPARK.items()}

FIXED_WIDTH_BYTES = {
    "bool": 1,
    "int8": 1,
    "int16": 2,
    "int32": 4,
    "int64": 8,
    "float16": 2,
    "float32": 4,
    "float64": 8,
    "date32": 4,
    "timestamp[us]": 8,
    "timestamp[ms]": 8,
    "timestamp[ns]": 8,
}


_DECIMAL_RE = None


def decimal_params(dtype: str):
    """Return (precision, scale) for a decimal dtype string, else None."""
    global _DECIMAL_RE
    if _DECIMAL_RE is None:
        import re
        _DECIMAL_RE = re.compile(r"decimal\s*\(\s*(\d+)\s*,\s*(\d+)\s*\)$")
    m = _DECIMAL_RE.match(dtype.strip().lower())
    return (int(m.group(1)), int(m.group(2))) if m else None


def list_element_dtype(dtype: str):
    """Return the element dtype of a list dtype string, else None."""
    s = dtype.strip().lower()
    for prefix in ("list<", "array<"):
        if s.startswith(prefix) and s.endswith(">"):
            return s[len(prefix):-1].strip()
    return None


def struct_members(dtype: str):
    """Parse ``struct<a:T1,b:T2,...>`` into [(name, dtype), ...], else
    None. Member dtypes may themselves contain commas (decimal(p,s));
    split at depth 0 of <> and () only."""
    s = dtype.strip()
    if not (s.lower().startswith("struct<") and s.endswith(">")):
        return None
    body = s[7:-1]
    parts, depth, cur = [], 0, []
    for ch in body:
        if ch in "<(":
            depth += 1
        elif ch in ">)":
            depth -= 1
        if ch == "," and depth == 0:
            parts.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    if cur:
        parts.append("".join(cur))
    out = []
    for p in parts:
        if ":" not in p:
            raise TypeError(f"bad struct member {p!r} in {dtype!r}")
        name, mdt = p.split(":", 1)
        out.append((name.strip(), mdt.strip()))
    if not out:
        raise TypeError(f"empty struct {dtype!r}")
    return out


def map_params(dtype: str):
    """Parse ``map<K,V>`` into (key_dtype, value_dtype), else None."""
    s = dtype.strip()
    if not (s.lower().startswith("map<") and s.endswith(">")):
        return None
    body = s[4:-1]
    depth, cur, parts = 0, [], []
    for ch in body:
        if ch in "<(":
            depth += 1
        elif ch in ">)":
            depth -= 1
        if ch == "," and depth == 0:
            parts.append("".join(cur))
            cur = []
        else:
            cur.append(ch)
    parts.append("".join(cur))
    if len(parts) != 2:
        raise TypeError(f"map needs exactly key,value: {dtype!r}")
    return parts[0].strip(), parts[1].strip()


def canonical_dtype(dt: str) -> str:
    sm = struct_members(dt)
    if sm is not None:
        mem = []
        for name, mdt in sm:
            inner = canonical_dtype(mdt)
            if inner not in FIXED_WIDTH_BYTES and inner != "string" \
                    and not inner.startswith("decimal"):
                raise TypeError(
                    f"unsupported struct member dtype {mdt!r} "
                    "(scalar members only)")
            mem.append(f"{name}:{inner}")
        return "struct<" + ",".join(mem) + ">"
    mp = map_params(dt)
    if mp is not None:
        k, v = canonical_dtype(mp[0]), canonical_dtype(mp[1])
        for inner in (k, v):
            if inner not in FIXED_WIDTH_BYTES and inner != "string":
                raise TypeError(
                    f"unsupported map key/value dtype {inner!r} "
                    "(scalar keys/values only)")
        return f"map<{k},{v}>"
    dp = decimal_params(dt)
    if dp is not None:
        p, sc = dp
        if not (0 < p <= 18) or not (0 <= sc <= p):
            raise TypeError(
                f"unsupported decimal precision/scale {dt} (int64-backed: p<=18)")
        return f"decimal({p},{sc})"
    elem = list_element_dtype(dt)
    if elem is not None:
        inner = canonical_dtype(elem)
        if inner not in FIXED_WIDTH_BYTES and inner != "string":
            raise TypeError(
                f"unsupported list element dtype {elem!r} "
                "(fixed-width primitives or string)")
        return f"list<{inner}>"
    try:
        return _CANONICAL[dt.lower()]
    except KeyError:
        raise TypeError(f"unsupported dtype: {dt}")


@dataclass
class Field:
    name: str
    dtype: str
    nullable: bool = True

    def __post_init__(self):
        self.dtype = canonical_dtype(self.dtype)

    @property
    def is_fixed_width(self) -> bool:
        return self.dtype in FIXED_WIDTH_BYTES or self.dtype.startswith("decimal")


@dataclass
class Schema:
    fields: List[Field] = dc_field(default_factory=list)

    def names(self) -> List[str]:
        return [f.name for f in self.fields]

    def field(self, name: str) -> Field:
        for f in self.fields:
            if f.name == name:
                return f
        raise KeyError(name)

    def index(self, name: str) -> int:
        for i, f in enumerate(self.fields):
            if f.name == name:
                return i
        raise KeyError(name)

    def select(self, names: Sequence[str]) -> "Schema":
        return Schema([self.field(n) for n in names])

    def __len__(self):
        return len(self.fields)

    def __iter__(self):
        return iter(self.fields)

    def __eq__(self, other):
        return isinstance(other, Schema) and self.fields == other.fields


SchemaLike = Union[Schema, Sequence[Tuple[str, str]], "object"]


def normalize_schema(s: SchemaLike) -> Schema:
    if isinstance(s, Schema):
        return s
    # pyarrow.Schema duck-typing
    if hasattr(s, "types") and hasattr(s, "names"):
        fields = []
        for f in s:  # type: ignore
            fields.append(Field(f.name, _arrow_type_to_dtype(f.type), f.nullable))
        return Schema(fields)
    # sequence of (name, dtype[, nullable])
    out = []
    for item in s:  # type: ignore
        if len(item) == 2:
            out.append(Field(item[0], item[1]))
        else:
            out.append(Field(item[0], item[1], item[2]))
    return Schema(out)


def _arrow_type_to_dtype(t) -> str:
    import pyarrow as pa

    mapping = {
        pa.bool_(): "bool",
        pa.int8(): "int8",
        pa.int16(): "int16",
        pa.int32(): "int32",
        pa.int64(): "int64",
        pa.float16(): "float16",
        pa.float32(): "float32",
        pa.float64(): "float64",
        pa.string(): "string",
        pa.large_string(): "string",
        pa.binary(): "binary",
        pa.large_binary(): "binary",
        pa.date32(): "date32",
    }
    if t in mapping:
        return mapping[t]
    import pyarrow.types as pt

    if pt.is_timestamp(t):
        return f"timestamp[{t.unit}]"
    if pt.is_decimal(t):
        return canonical_dtype(f"decimal({t.precision},{t.scale})")
    if pt.is_list(t) or pt.is_large_list(t):
        inner = _arrow_type_to_dtype(t.value_type)
        return canonical_dtype(f"list<{inner}>")
    if pt.is_struct(t):
        mem = ",".join(f"{t.field(i).name}:{_arrow_type_to_dtype(t.field(i).type)}"
                       for i in range(t.num_fields))
        return canonical_dtype(f"struct<{mem}>")
    if pt.is_map(t):
        return canonical_dtype(
            f"map<{_arrow_type_to_dtype(t.key_type)},"
            f"{_arrow_type_to_dtype(t.item_type)}>")
    raise TypeError(f"unsupported arrow type {t}")


def dtype_to_arrow(dtype: str):
    import pyarrow as pa

    mapping = {
        "bool": pa.bool_(),
        "int8": pa.int8(),
        "int16": pa.int16(),
        "int32": pa.int32(),
        "int64": pa.int64(),
        "float16": pa.float16(),
        "float32": pa.float32(),
        "float64": pa.float64(),
        "string": pa.string(),
        "binary": pa.binary(),
        "date32": pa.date32(),
        "timestamp[us]": pa.timestamp("us"),
        "timestamp[ms]": pa.timestamp("ms"),
        "timestamp[ns]": pa.timestamp("ns"),
    }
    dp = decimal_params(dtype)
    if dp is not None:
        return pa.decimal128(*dp)
    elem = list_element_dtype(dtype)
    if elem is not None:
        return pa.list_(dtype_to_arrow(elem))
    sm = struct_members(dtype)
    if sm is not None:
        return pa.struct([pa.field(n, dtype_to_arrow(t), nullable=False)
                          for n, t in sm])
    mp = map_params(dtype)
    if mp is not None:
        return pa.map_(dtype_to_arrow(mp[0]), dtype_to_arrow(mp[1]))
    return mapping[dtype]


def schema_to_arrow(schema: Schema):
    import pyarrow as pa

    return pa.schema(
        [pa.field(f.name, dtype_to_arrow(f.dtype), nullable=f.nullable) for f in schema]
    )


def _dtype_to_spark(dtype: str):
    elem = list_element_dtype(dtype)
    if elem is not None:
        # spark array type JSON (reference ser/arrow_java.rs array handling)
        return {"type": "array",
                "elementType": _TO_SPARK.get(elem, elem),
                "containsNull": False}
    sm = struct_members(dtype)
    if sm is not None:
        return {"type": "struct",
                "fields": [{"name": n, "type": _dtype_to_spark(t),
                            "nullable": False, "metadata": {}}
                           for n, t in sm]}
    mp = map_params(dtype)
    if mp is not None:
        return {"type": "map",
                "keyType": _TO_SPARK.get(mp[0], mp[0]),
                "valueType": _TO_SPARK.get(mp[1], mp[1]),
                "valueContainsNull": False}
    return _TO_SPARK.get(dtype, dtype)


def _dtype_from_spark(t):
    if isinstance(t, dict) and t.get("type") == "array":
        et = t.get("elementType")
        return f"list<{_FROM_SPARK.get(et, et)}>"
    if isinstance(t, dict) and t.get("type") == "struct":
        mem = ",".join(
            f"{f['name']}:{_dtype_from_spark(f['type'])}"
            for f in t.get("fields", []))
        return f"struct<{mem}>"
    if isinstance(t, dict) and t.get("type") == "map":
        kt, vt = t.get("keyType"), t.get("valueType")
        return (f"map<{_FROM_SPARK.get(kt, kt)},"
                f"{_FROM_SPARK.get(vt, vt)}>")
    return _FROM_SPARK.get(t, t)


def schema_to_json(schema: Schema) -> str:
    return json.dumps(
        {
            "type": "struct",
            "fields": [
                {
                    "name": f.name,
                    "type": _dtype_to_spark(f.dtype),
                    "nullable": f.nullable,
                    "metadata": {},
                }
                for f in schema
            ],
        }
    )


def schema_from_json(s: str) -> Schema:
    d = json.loads(s)
    fields = []
    for f in d.get("fields", []):
        fields.append(Field(f["name"], _dtype_from_spark(f["type"]),
                            bool(f.get("nullable", True))))
    return Schema(fields)
