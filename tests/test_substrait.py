"""Substrait filter intake (reference: rust/lakesoul-io/src/filter/
parser.rs:44-60 FilterContainer {RawBuf, Plan, ExtendedExpression}).

Three layers of evidence:
1. round-trip through our own encoder (ExtendedExpression AND Plan);
2. decode of REAL bytes produced by an independent producer
   (pyarrow.substrait / Acero serialize_expressions);
3. DSL-equivalence: identical pruning + row-selection decisions for
   equivalent DSL and Substrait inputs (the VERDICT done-criterion).
"""

import numpy as np
import pytest

from lakesoul_amd.io.batch import Batch
from lakesoul_amd.io.filters import (
    And, Cmp, IsNull, Literal, Not, Or, parse_filter_dsl, resolve_filters)
from lakesoul_amd.io.schema import Field, Schema
from lakesoul_amd.io.substrait import (
    SubstraitError, decode_substrait_filter, encode_substrait_filter,
    encode_substrait_plan_filter)

SCHEMA = Schema([
    Field("id", "int64", False),
    Field("v", "float64"),
    Field("s", "string"),
    Field("flag", "bool"),
])


def _batch(n=100):
    return Batch.from_dict({
        "id": np.arange(n, dtype=np.int64),
        "v": np.linspace(0, 1, n),
        "s": [None if i % 10 == 0 else f"s{i % 7}" for i in range(n)],
        "flag": np.array([i % 2 == 0 for i in range(n)]),
    }, SCHEMA)


EXPRS = [
    Cmp("id", "gt", 10),
    Cmp("v", "lteq", 0.5),
    Cmp("s", "eq", "s3"),
    And(Cmp("id", "gteq", 5), Or(Cmp("v", "lt", 0.9), Cmp("s", "noteq", "s1"))),
    Not(Cmp("id", "eq", 42)),
    IsNull("s"),
    IsNull("s", negate=True),
    Cmp("id", "in", [1, 5, 9, 77]),
    And(IsNull("s", negate=True), Cmp("flag", "eq", True)),
]


@pytest.mark.parametrize("expr", EXPRS, ids=[str(i) for i in range(len(EXPRS))])
def test_roundtrip_extended_expression(expr):
    buf = encode_substrait_filter(expr, SCHEMA)
    got = decode_substrait_filter(buf, SCHEMA)
    b = _batch()
    np.testing.assert_array_equal(
        got.evaluate(b).numpy(), expr.evaluate(b).numpy())


@pytest.mark.parametrize("expr", EXPRS[:5], ids=[str(i) for i in range(5)])
def test_roundtrip_plan(expr):
    buf = encode_substrait_plan_filter(expr, SCHEMA)
    got = decode_substrait_filter(buf, SCHEMA)
    b = _batch()
    np.testing.assert_array_equal(
        got.evaluate(b).numpy(), expr.evaluate(b).numpy())


def _pa_bytes(expr_pc, schema_pa):
    import pyarrow.substrait as ps

    return bytes(memoryview(ps.serialize_expressions([expr_pc], ["f"], schema_pa)))


def test_decode_pyarrow_produced_bytes():
    """Independent-producer check: Acero's serializer, our decoder."""
    pa = pytest.importorskip("pyarrow")
    pc = pytest.importorskip("pyarrow.compute")
    pytest.importorskip("pyarrow.substrait")
    schema_pa = pa.schema([("id", pa.int64()), ("v", pa.float64()), ("s", pa.string())])

    cases = [
        ((pc.field("id") > 10), Cmp("id", "gt", 10)),
        ((pc.field("id") <= 3), Cmp("id", "lteq", 3)),
        ((pc.field("s") == "x"), Cmp("s", "eq", "x")),
        ((pc.field("id") > 10) & (pc.field("s") == "x"),
         And(Cmp("id", "gt", 10), Cmp("s", "eq", "x"))),
        ((pc.field("v") < 0.25) | (pc.field("id") != 7),
         Or(Cmp("v", "lt", 0.25), Cmp("id", "noteq", 7))),
        (pc.field("s").is_null(), IsNull("s")),
        (~(pc.field("id") == 5), Not(Cmp("id", "eq", 5))),
    ]
    b = _batch()
    for expr_pc, expect in cases:
        got = decode_substrait_filter(_pa_bytes(expr_pc, schema_pa), SCHEMA)
        np.testing.assert_array_equal(
            got.evaluate(b).numpy(), expect.evaluate(b).numpy(),
            err_msg=str(expr_pc))


DSL_EQUIV = [
    ("gt(id, 10)", Cmp("id", "gt", 10)),
    ("and(gteq(id, 5), lt(v, 0.9))", And(Cmp("id", "gteq", 5), Cmp("v", "lt", 0.9))),
    ("or(eq(s, 's3'), noteq(id, 7))", Or(Cmp("s", "eq", "s3"), Cmp("id", "noteq", 7))),
    ("not(eq(id, 42))", Not(Cmp("id", "eq", 42))),
    ("eq(s, null)", IsNull("s")),
    ("noteq(s, null)", IsNull("s", negate=True)),
]


@pytest.mark.parametrize("dsl,ir", DSL_EQUIV, ids=[d for d, _ in DSL_EQUIV])
def test_dsl_substrait_equivalence(dsl, ir):
    """Equivalent DSL and Substrait filters make identical decisions:
    same rows selected, same stats-pruning verdicts, same partition
    pruning, same pk-eq extraction."""
    from_dsl = parse_filter_dsl(dsl, SCHEMA)
    from_sub = decode_substrait_filter(encode_substrait_filter(ir, SCHEMA), SCHEMA)
    b = _batch()
    np.testing.assert_array_equal(
        from_dsl.evaluate(b).numpy(), from_sub.evaluate(b).numpy())
    # stats pruning decisions
    for stats in [
        {"id": (0, 4), "v": (0.95, 1.0), "s": ("a", "z")},
        {"id": (11, 99), "v": (0.0, 0.5), "s": ("s3", "s3")},
        {"id": (42, 42), "v": (0.9, 1.0), "s": ("s0", "s9")},
    ]:
        assert from_dsl.prune_stats(stats) == from_sub.prune_stats(stats)
    assert from_dsl.pk_eq_values() == from_sub.pk_eq_values()
    for pv in [{"s": "s3"}, {"s": None}, {"id": "42"}, {}]:
        assert from_dsl.partition_prune(pv) == from_sub.partition_prune(pv)


def test_scan_accepts_substrait_bytes(catalog):
    """End to end: table.scan(filters=<substrait bytes>) prunes and
    filters like the tuple/DSL input."""
    t = catalog.create_table(
        "subst",
        Schema([Field("id", "int64", False), Field("v", "float64")]),
        primary_keys=["id"], hash_bucket_num=2,
    )
    t.upsert({"id": np.arange(1000, dtype=np.int64),
              "v": np.arange(1000, dtype=np.float64)})
    buf = encode_substrait_filter(
        And(Cmp("id", "gteq", 100), Cmp("id", "lt", 110)), t.schema)
    df = t.scan(filters=buf).to_arrow().to_pandas().sort_values("id")
    assert df["id"].tolist() == list(range(100, 110))
    # point filter via substrait engages bucket pruning like the tuple path
    buf_pt = encode_substrait_filter(Cmp("id", "eq", 123), t.schema)
    scan_sub = t.scan(filters=buf_pt)
    scan_tup = t.scan(filters=[("id", "==", 123)])
    assert len(scan_sub.plan()) == len(scan_tup.plan())
    df2 = scan_sub.to_arrow().to_pandas()
    assert df2["id"].tolist() == [123]


def test_unsupported_raises():
    with pytest.raises((SubstraitError, TypeError)):
        decode_substrait_filter(b"\x00garbage\xff\xff", SCHEMA)
    with pytest.raises(SubstraitError):
        decode_substrait_filter(b"", SCHEMA)
