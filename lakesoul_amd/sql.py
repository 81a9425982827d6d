"""Minimal SQL layer over the catalog — the MI355X-native stand-in for
the reference's ``lakesoul-console`` interactive SQL REPL
(``rust/lakesoul-console/src/exec.rs``) and the datafusion CLI
(``rust/lakesoul-datafusion/src/cli.rs``).

The reference delegates SQL to DataFusion; here a hand-rolled
recursive-descent parser covers the console's practical surface —
``SELECT`` with projection, aggregates, ``WHERE`` (pushed down into the
scan's filter/stats/bucket pruning), ``GROUP BY``, ``ORDER BY``,
``LIMIT``, plus ``SHOW TABLES / NAMESPACES`` and ``DESCRIBE`` — and
executes against :class:`~lakesoul_amd.tables.catalog.Catalog` scans, so
queries get the same MOR + pruning machinery as the programmatic API.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from .io.filters import And, Cmp, Expr, IsNull, Literal, Not, Or

_TOKEN_RE = re.compile(
    r"""\s*(?:
        (?P<num>\d+\.\d+(?:[eE][+-]?\d+)?|\.\d+|\d+)
      | (?P<str>'(?:[^']|'')*')
      | (?P<op><=|>=|<>|!=|=|<|>|\(|\)|,|\*|\.|\+|-|/|:)
      | (?P<id>[A-Za-z_][A-Za-z_0-9]*)
    )""",
    re.VERBOSE,
)

_KEYWORDS = {
    "select", "from", "where", "group", "order", "by", "limit", "and",
    "or", "not", "in", "is", "null", "between", "as", "asc", "desc",
    "show", "tables", "namespaces", "describe", "distinct", "version",
    "join", "inner", "left", "on", "insert", "into", "values",
    "update", "set", "delete", "offset", "having", "timestamp", "explain",
    "create", "table", "drop", "primary", "key", "hash", "buckets",
    "partition", "if", "exists", "alter", "add", "column", "compact",
    "vacuum", "keep",
}

_AGGS = {"count", "sum", "min", "max", "avg"}


class SqlError(ValueError):
    pass


@dataclass
class SubqueryIn(Expr):
    """col IN (SELECT ...) — non-correlated; resolved to an OR-of-eq /
    isin before execution."""
    col: str
    query: "Query"

    def evaluate(self, batch):  # pragma: no cover - resolved earlier
        raise SqlError("unresolved subquery")


@dataclass
class SubqueryCmp(Expr):
    """col <op> (SELECT scalar ...) — resolved to a Cmp literal."""
    col: str
    op: str
    query: "Query"

    def evaluate(self, batch):  # pragma: no cover - resolved earlier
        raise SqlError("unresolved subquery")


def tokenize(sql: str) -> List[Tuple[str, str]]:
    out, pos = [], 0
    sql = sql.strip().rstrip(";")
    while pos < len(sql):
        m = _TOKEN_RE.match(sql, pos)
        if not m:
            raise SqlError(f"bad token at: {sql[pos:pos+20]!r}")
        pos = m.end()
        if m.lastgroup == "num":
            out.append(("num", m.group("num")))
        elif m.lastgroup == "str":
            out.append(("str", m.group("str")[1:-1].replace("''", "'")))
        elif m.lastgroup == "op":
            out.append(("op", m.group("op")))
        else:
            word = m.group("id")
            kind = "kw" if word.lower() in _KEYWORDS else "id"
            out.append((kind, word.lower() if kind == "kw" else word))
    return out


# --------------------------------------------------------------------- #
# AST


@dataclass
class Scalar:
    """Arithmetic scalar expression over columns and literals (the
    reference gets this from DataFusion's ProjectionExec;
    session.rs:966-1036)."""
    kind: str                   # "col" | "num" | "bin"
    name: str = ""              # col
    value: float = 0.0          # num
    op: str = ""                # + - * /
    left: Optional["Scalar"] = None
    right: Optional["Scalar"] = None

    def columns(self) -> set:
        if self.kind == "col":
            return {self.name}
        if self.kind == "bin":
            return self.left.columns() | self.right.columns()
        return set()

    def display(self) -> str:
        if self.kind == "col":
            return self.name
        if self.kind == "num":
            v = self.value
            return str(int(v)) if float(v).is_integer() else str(v)
        return f"({self.left.display()} {self.op} {self.right.display()})"


@dataclass
class SelectItem:
    # one of: column name | ("agg", fn, col_or_star) | ("lit", value)
    kind: str                      # "col" | "agg" | "star"
    name: str = ""                 # column name (col/agg arg)
    fn: str = ""                   # aggregate fn
    alias: str = ""
    distinct: bool = False         # count(DISTINCT col)
    expr: Optional[Scalar] = None  # computed scalar (name empty)

    @property
    def out_name(self) -> str:
        if self.alias:
            return self.alias
        if self.kind == "agg":
            arg = self.expr.display() if self.expr is not None else (self.name or "*")
            return f"{self.fn}({arg})"
        if self.expr is not None:
            return self.expr.display()
        return self.name


@dataclass
class JoinSpec:
    table: str
    namespace: str = "default"
    alias: str = ""
    kind: str = "inner"               # inner | left
    on: List[Tuple[str, str]] = field(default_factory=list)  # (left col, right col)


@dataclass
class Query:
    table: str
    namespace: str = "default"
    alias: str = ""
    joins: List["JoinSpec"] = field(default_factory=list)
    items: List[SelectItem] = field(default_factory=list)
    where: Optional[Expr] = None
    group_by: List[str] = field(default_factory=list)
    order_by: List[Tuple[str, bool]] = field(default_factory=list)  # (name, desc)
    having: Optional[Expr] = None
    limit: Optional[int] = None
    offset: int = 0
    distinct: bool = False
    version: Optional[int] = None
    timestamp_ms: Optional[int] = None


class _Parser:
    def __init__(self, tokens):
        self.t = tokens
        self.i = 0

    def peek(self, k=0):
        j = self.i + k
        return self.t[j] if j < len(self.t) else ("eof", "")

    def next(self):
        tok = self.peek()
        self.i += 1
        return tok

    def expect(self, kind, val=None):
        k, v = self.next()
        if k != kind or (val is not None and v != val):
            raise SqlError(f"expected {val or kind}, got {v!r}")
        return v

    def accept(self, kind, val=None):
        k, v = self.peek()
        if k == kind and (val is None or v == val):
            self.i += 1
            return True
        return False

    # -- statements ----------------------------------------------------- #

    def statement(self):
        k, v = self.peek()
        if (k, v) == ("kw", "show"):
            self.next()
            k2, v2 = self.next()
            if (k2, v2) == ("kw", "tables") or (k2, v2) == ("kw", "namespaces"):
                return ("show", v2)
            if k2 == "id" and v2.lower() in ("partitions", "history"):
                ns, name = self.table_name()
                return ("show_" + v2.lower(), {"namespace": ns, "table": name})
            if (k2, v2) == ("kw", "create"):
                self.expect("kw", "table")
                ns, name = self.table_name()
                return ("show_create", {"namespace": ns, "table": name})
            raise SqlError(f"SHOW {v2}?")
        if (k, v) == ("kw", "describe"):
            self.next()
            ns, name = self.table_name()
            return ("describe", (ns, name))
        if (k, v) == ("kw", "explain"):
            self.next()
            return ("explain", self.select())
        if (k, v) == ("kw", "select"):
            return ("select", self.select())
        if (k, v) == ("kw", "insert"):
            return ("insert", self.insert())
        if (k, v) == ("kw", "create"):
            return ("create", self.create_table())
        if (k, v) == ("kw", "compact"):
            self.next()
            self.accept("kw", "table")
            ns, name = self.table_name()
            return ("compact", {"namespace": ns, "table": name})
        if (k, v) == ("kw", "vacuum"):
            self.next()
            ns, name = self.table_name()
            keep = 1
            if self.accept("kw", "keep"):
                keep = int(self.expect("num"))
            return ("vacuum", {"namespace": ns, "table": name, "keep": keep})
        if (k, v) == ("kw", "alter"):
            self.next()
            self.expect("kw", "table")
            ns, name = self.table_name()
            self.expect("kw", "add")
            self.accept("kw", "column")
            cols = []
            while True:
                cname = self.expect("id")
                dtype = self.sql_type()
                cols.append((cname, dtype))
                if not self.accept("op", ","):
                    break
            return ("alter_add", {"namespace": ns, "table": name,
                                  "columns": cols})
        if (k, v) == ("kw", "drop"):
            self.next()
            self.expect("kw", "table")
            if_exists = False
            if self.accept("kw", "if"):
                self.expect("kw", "exists")
                if_exists = True
            ns, name = self.table_name()
            return ("drop", {"namespace": ns, "table": name,
                             "if_exists": if_exists})
        if (k, v) == ("kw", "update"):
            self.next()
            ns, name = self.table_name()
            self.expect("kw", "set")
            assigns = {}
            while True:
                col = self.expect("id")
                self.expect("op", "=")
                assigns[col] = self.literal()
                if not self.accept("op", ","):
                    break
            where = None
            if self.accept("kw", "where"):
                where = self.or_expr()
            return ("update", {"namespace": ns, "table": name,
                               "assignments": assigns, "where": where})
        if (k, v) == ("kw", "delete"):
            self.next()
            self.expect("kw", "from")
            ns, name = self.table_name()
            where = None
            if self.accept("kw", "where"):
                where = self.or_expr()
            return ("delete", {"namespace": ns, "table": name, "where": where})
        raise SqlError(f"unsupported statement start: {v!r}")

    def insert(self):
        self.expect("kw", "insert")
        self.expect("kw", "into")
        ns, name = self.table_name()
        cols = []
        if self.accept("op", "("):
            cols.append(self.expect("id"))
            while self.accept("op", ","):
                cols.append(self.expect("id"))
            self.expect("op", ")")
        k, v = self.peek()
        if (k, v) == ("kw", "values"):
            self.next()
            rows = []
            while True:
                self.expect("op", "(")
                row = [self.literal()]
                while self.accept("op", ","):
                    row.append(self.literal())
                self.expect("op", ")")
                rows.append(row)
                if not self.accept("op", ","):
                    break
            return {"namespace": ns, "table": name, "columns": cols, "rows": rows}
        if (k, v) == ("kw", "select"):
            return {"namespace": ns, "table": name, "columns": cols,
                    "select": self.select()}
        raise SqlError("INSERT expects VALUES or SELECT")

    _SQL_TYPES = {
        "bigint": "int64", "long": "int64", "int": "int32",
        "integer": "int32", "smallint": "int16", "tinyint": "int8",
        "double": "float64", "float": "float32", "real": "float32",
        "varchar": "string", "string": "string", "text": "string",
        "binary": "binary", "boolean": "bool", "bool": "bool",
        "date": "date32", "bigint_ts": "timestamp[us]",
    }

    def sql_type(self) -> str:
        t = self.expect("id").lower()
        if t == "decimal" and self.accept("op", "("):
            p_ = self.expect("num")
            self.expect("op", ",")
            sc = self.expect("num")
            self.expect("op", ")")
            return f"decimal({p_},{sc})"
        if t == "timestamp":
            return "timestamp[us]"
        if t == "varchar" and self.accept("op", "("):
            self.expect("num")
            self.expect("op", ")")
            return "string"
        if t in self._SQL_TYPES:
            return self._SQL_TYPES[t]
        # canonical engine dtype names round-trip (SHOW CREATE TABLE
        # emits them; nested types like struct<...> arrive as one token
        # stream — re-join the <...> args)
        if t in ("struct", "map", "list", "array") and self.accept("op", "<"):
            depth, parts = 1, [t, "<"]
            while depth > 0:
                k, v = self.next()
                if (k, v) == ("op", "<"):
                    depth += 1
                elif (k, v) == ("op", ">"):
                    depth -= 1
                parts.append(v if k != "str" else f"'{v}'")
            t = "".join(parts)
        try:
            from .io.schema import canonical_dtype

            return canonical_dtype(t)
        except TypeError:
            raise SqlError(f"unknown SQL type {t!r}")

    def create_table(self):
        self.expect("kw", "create")
        self.expect("kw", "table")
        ns, name = self.table_name()
        self.expect("op", "(")
        cols = []
        while True:
            cname = self.expect("id")
            dtype = self.sql_type()
            nullable = True
            if self.accept("kw", "not"):
                self.expect("kw", "null")
                nullable = False
            cols.append((cname, dtype, nullable))
            if not self.accept("op", ","):
                break
        self.expect("op", ")")
        pks, buckets, parts = [], 1, []
        while self.peek()[0] != "eof":
            if self.accept("kw", "primary"):
                self.expect("kw", "key")
                self.expect("op", "(")
                pks.append(self.expect("id"))
                while self.accept("op", ","):
                    pks.append(self.expect("id"))
                self.expect("op", ")")
            elif self.accept("kw", "hash"):
                self.expect("kw", "buckets")
                buckets = int(self.expect("num"))
            elif self.accept("kw", "partition"):
                self.expect("kw", "by")
                self.expect("op", "(")
                parts.append(self.expect("id"))
                while self.accept("op", ","):
                    parts.append(self.expect("id"))
                self.expect("op", ")")
            else:
                raise SqlError(f"unexpected token {self.peek()[1]!r}")
        return {"namespace": ns, "table": name, "columns": cols,
                "primary_keys": pks, "hash_buckets": buckets,
                "range_partitions": parts}

    def table_name(self):
        name = self.expect("id")
        if self.accept("op", "."):
            return name, self.expect("id")
        return "default", name

    def qualified_id(self) -> str:
        name = self.expect("id")
        if self.accept("op", "."):
            return f"{name}.{self.expect('id')}"
        return name

    # -- SELECT ---------------------------------------------------------- #

    def select(self, nested: bool = False) -> Query:
        self.expect("kw", "select")
        q = Query(table="")
        q.distinct = self.accept("kw", "distinct")
        q.items = [self.select_item()]
        while self.accept("op", ","):
            q.items.append(self.select_item())
        self.expect("kw", "from")
        q.namespace, q.table = self.table_name()
        if self.peek()[0] == "id":
            q.alias = self.next()[1]
        if self.accept("kw", "version"):   # time travel: FROM t VERSION 3
            q.version = int(self.expect("num"))
        elif self.accept("kw", "timestamp"):  # FROM t TIMESTAMP <epoch_ms>
            q.timestamp_ms = int(self.expect("num"))
        while True:
            k, v = self.peek()
            if not (k == "kw" and v in ("join", "inner", "left")):
                break
            kind = "inner"
            if self.accept("kw", "left"):
                kind = "left"
            else:
                self.accept("kw", "inner")
            self.expect("kw", "join")
            jns, jname = self.table_name()
            j = JoinSpec(jname, jns, kind=kind)
            if self.peek()[0] == "id":
                j.alias = self.next()[1]
            self.expect("kw", "on")
            while True:
                lcol = self.qualified_id()
                self.expect("op", "=")
                rcol = self.qualified_id()
                j.on.append((lcol, rcol))
                if not self.accept("kw", "and"):
                    break
            q.joins.append(j)
        if self.accept("kw", "where"):
            q.where = self.or_expr()
        if self.accept("kw", "group"):
            self.expect("kw", "by")
            q.group_by = [self.qualified_id()]
            while self.accept("op", ","):
                q.group_by.append(self.qualified_id())
        if self.accept("kw", "having"):
            q.having = self.or_expr()
        if self.accept("kw", "order"):
            self.expect("kw", "by")
            q.order_by = [self.order_item(q)]
            while self.accept("op", ","):
                q.order_by.append(self.order_item(q))
        if self.accept("kw", "limit"):
            q.limit = int(self.expect("num"))
        if self.accept("kw", "offset"):
            q.offset = int(self.expect("num"))
        if not nested and self.peek()[0] != "eof":
            raise SqlError(f"trailing tokens: {self.peek()[1]!r}")
        return q

    # ---- scalar expressions (q1/q6-style arithmetic) ---- #

    def scalar_expr(self) -> Scalar:
        e = self.scalar_term()
        while self.peek() in (("op", "+"), ("op", "-")):
            op = self.next()[1]
            e = Scalar("bin", op=op, left=e, right=self.scalar_term())
        return e

    def scalar_term(self) -> Scalar:
        e = self.scalar_factor()
        while self.peek() in (("op", "*"), ("op", "/")):
            op = self.next()[1]
            e = Scalar("bin", op=op, left=e, right=self.scalar_factor())
        return e

    def scalar_factor(self) -> Scalar:
        k, v = self.peek()
        if (k, v) == ("op", "("):
            self.next()
            e = self.scalar_expr()
            self.expect("op", ")")
            return e
        if (k, v) == ("op", "-"):
            self.next()
            inner = self.scalar_factor()
            return Scalar("bin", op="-", left=Scalar("num", value=0.0),
                          right=inner)
        if k == "num":
            self.next()
            return Scalar("num", value=float(v))
        if k == "id":
            return Scalar("col", name=self.qualified_id())
        raise SqlError(f"bad scalar expression near {v!r}")

    def _maybe_scalar(self) -> Tuple[str, Optional[Scalar]]:
        """Parse a scalar expression; pure column refs collapse to a
        plain name (backward-compatible fast path)."""
        e = self.scalar_expr()
        if e.kind == "col":
            return e.name, None
        return "", e

    def select_item(self) -> SelectItem:
        k, v = self.peek()
        if (k, v) == ("op", "*"):
            self.next()
            return SelectItem("star")
        if k == "id" and v.lower() in _AGGS and self.peek(1) == ("op", "("):
            fn = self.next()[1].lower()
            self.expect("op", "(")
            distinct = self.accept("kw", "distinct")
            if self.accept("op", "*"):
                if fn != "count":
                    raise SqlError(f"{fn}(*) not supported")
                arg, expr = "", None
            else:
                arg, expr = self._maybe_scalar()
            self.expect("op", ")")
            if distinct and fn != "count":
                raise SqlError("DISTINCT only supported inside count()")
            item = SelectItem("agg", name=arg, fn=fn, distinct=distinct,
                              expr=expr)
        elif k in ("id", "num") or (k, v) in (("op", "("), ("op", "-")):
            arg, expr = self._maybe_scalar()
            item = SelectItem("col", name=arg, expr=expr)
        else:
            raise SqlError(f"bad select item near {v!r}")
        if self.accept("kw", "as"):
            item.alias = self.expect("id")
        elif self.peek()[0] == "id":   # bare alias
            item.alias = self.next()[1]
        return item

    def order_item(self, q: Query):
        name = self.qualified_id() if self.peek()[0] == "id" else str(self.expect("num"))
        if name.isdigit():
            name = q.items[int(name) - 1].out_name
        desc = False
        if self.accept("kw", "desc"):
            desc = True
        else:
            self.accept("kw", "asc")
        return name, desc

    # -- WHERE ----------------------------------------------------------- #

    def or_expr(self) -> Expr:
        e = self.and_expr()
        while self.accept("kw", "or"):
            e = Or(e, self.and_expr())
        return e

    def and_expr(self) -> Expr:
        e = self.not_expr()
        while self.accept("kw", "and"):
            e = And(e, self.not_expr())
        return e

    def not_expr(self) -> Expr:
        if self.accept("kw", "not"):
            return Not(self.not_expr())
        return self.predicate()

    def literal(self):
        k, v = self.next()
        if (k, v) == ("op", "-"):
            k2, v2 = self.next()
            if k2 != "num":
                raise SqlError(f"expected number after '-', got {v2!r}")
            val = float(v2) if ("." in v2 or "e" in v2.lower()) else int(v2)
            return -val
        if k == "num":
            return float(v) if ("." in v or "e" in v.lower()) else int(v)
        if k == "str":
            return v
        if k == "id" and v.lower() in ("true", "false"):
            return v.lower() == "true"
        if k == "kw" and v == "null":
            return None
        raise SqlError(f"expected literal, got {v!r}")

    def predicate(self) -> Expr:
        if self.accept("op", "("):
            e = self.or_expr()
            self.expect("op", ")")
            return e
        col = self.qualified_id()
        k, v = self.peek()
        if (k, v) == ("kw", "is"):
            self.next()
            neg = self.accept("kw", "not")
            self.expect("kw", "null")
            return IsNull(col, negate=neg)
        if (k, v) == ("kw", "in") or ((k, v) == ("kw", "not") and self.peek(1) == ("kw", "in")):
            neg = self.accept("kw", "not")
            self.expect("kw", "in")
            self.expect("op", "(")
            if self.peek() == ("kw", "select"):
                sub = self.select(nested=True)
                self.expect("op", ")")
                e = SubqueryIn(col, sub)
                return Not(e) if neg else e
            vals = [self.literal()]
            while self.accept("op", ","):
                vals.append(self.literal())
            self.expect("op", ")")
            e: Expr = Cmp(col, "eq", vals[0])
            for x in vals[1:]:
                e = Or(e, Cmp(col, "eq", x))
            return Not(e) if neg else e
        if (k, v) == ("kw", "between") or ((k, v) == ("kw", "not") and self.peek(1) == ("kw", "between")):
            neg = self.accept("kw", "not")
            self.expect("kw", "between")
            lo = self.literal()
            self.expect("kw", "and")
            hi = self.literal()
            e = And(Cmp(col, "gteq", lo), Cmp(col, "lteq", hi))
            return Not(e) if neg else e
        op = self.expect("op")
        ops = {"=": "eq", "!=": "noteq", "<>": "noteq", "<": "lt",
               "<=": "lteq", ">": "gt", ">=": "gteq"}
        if op not in ops:
            raise SqlError(f"bad comparison operator {op!r}")
        if self.peek() == ("op", "(") and self.peek(1) == ("kw", "select"):
            self.next()
            sub = self.select(nested=True)
            self.expect("op", ")")
            return SubqueryCmp(col, ops[op], sub)
        return Cmp(col, ops[op], self.literal())


def parse_sql(sql: str):
    return _Parser(tokenize(sql)).statement()


# --------------------------------------------------------------------- #
# Execution


def execute_sql(catalog, sql: str, device: Optional[str] = None):
    """Run a SQL statement and return a pandas DataFrame."""
    import pandas as pd

    kind, payload = parse_sql(sql)
    if kind == "insert":
        return _execute_insert(catalog, payload, device=device)
    if kind == "create":
        import pandas as pd

        from .io.schema import Field, Schema

        t = catalog.create_table(
            payload["table"],
            Schema([Field(n, d, nu) for n, d, nu in payload["columns"]]),
            primary_keys=payload["primary_keys"],
            hash_bucket_num=payload["hash_buckets"],
            range_partitions=payload["range_partitions"],
            namespace=payload["namespace"],
        )
        return pd.DataFrame({"table_id": [t.table_id]})
    if kind == "show_partitions":
        import pandas as pd

        t = catalog.table(payload["table"], payload["namespace"])
        descs = t.partition_descs()
        return pd.DataFrame({
            "partition": descs,
            "latest_version": [t.latest_version(d) for d in descs],
        })
    if kind == "show_history":
        import pandas as pd

        t = catalog.table(payload["table"], payload["namespace"])
        rows = []
        for desc in t.partition_descs():
            cur = t.client.store.get_latest_partition_info(t.table_id, desc)
            if cur is None:
                continue
            for p_ in t.client.store.get_partition_versions_in_range(
                    t.table_id, desc, 0, cur.version):
                rows.append({"partition": desc, "version": p_.version,
                             "commit_op": p_.commit_op.name,
                             "timestamp_ms": p_.timestamp,
                             "commits": len(p_.snapshot)})
        return pd.DataFrame(rows)
    if kind == "compact":
        import pandas as pd

        t = catalog.table(payload["table"], payload["namespace"])
        t.compaction()
        return pd.DataFrame({"compacted": [payload["table"]]})
    if kind == "vacuum":
        import pandas as pd

        t = catalog.table(payload["table"], payload["namespace"])
        removed = t.cleanup_old_versions(keep_latest=payload["keep"])
        return pd.DataFrame({"files_removed": [removed]})
    if kind == "alter_add":
        import pandas as pd

        from .io.schema import Field

        t = catalog.table(payload["table"], payload["namespace"])
        t.add_columns([Field(n, d) for n, d in payload["columns"]])
        return pd.DataFrame({"added": [len(payload["columns"])]})
    if kind == "drop":
        import pandas as pd

        try:
            catalog.drop_table(payload["table"], payload["namespace"],
                               delete_data=True)
        except Exception:
            if not payload["if_exists"]:
                raise
        return pd.DataFrame({"dropped": [payload["table"]]})
    if kind == "explain":
        return _explain_select(catalog, payload, device=device)
    if kind in ("update", "delete"):
        import pandas as pd

        t = catalog.table(payload["table"], payload["namespace"])
        if kind == "update":
            n = t.update(payload["where"], payload["assignments"], device=device)
            return pd.DataFrame({"rows_updated": [n]})
        n = t.delete(payload["where"], device=device)
        return pd.DataFrame({"rows_deleted": [n]})
    if kind == "show":
        if payload == "namespaces":
            return pd.DataFrame({"namespace": catalog.list_namespaces()})
        return pd.DataFrame({"table": catalog.list_tables()})
    if kind == "show_create":
        t = catalog.table(payload["table"], payload["namespace"])
        sch = t.schema
        cols = ", ".join(
            f"{f.name} {f.dtype.upper() if f.dtype in ('string',) else f.dtype}"
            + ("" if f.nullable else " NOT NULL") for f in sch)
        stmt = f"CREATE TABLE {payload['table']} ({cols})"
        if t.primary_keys:
            stmt += f" PRIMARY KEY ({', '.join(t.primary_keys)})"
        stmt += f" HASH BUCKETS {t.hash_bucket_num}"
        if t.range_keys:
            stmt += f" PARTITION BY ({', '.join(t.range_keys)})"
        return pd.DataFrame({"table": [payload["table"]],
                             "create_statement": [stmt]})
    if kind == "describe":
        ns, name = payload
        t = catalog.table(name, ns)
        sch = t.schema
        return pd.DataFrame({
            "column": [f.name for f in sch],
            "type": [f.dtype for f in sch],
            "nullable": [f.nullable for f in sch],
            "primary_key": [f.name in (t.primary_keys or []) for f in sch],
        })
    return _execute_select(catalog, payload, device=device)


def _is_null_cell(v) -> bool:
    if v is None:
        return True
    try:
        import math

        return isinstance(v, float) and math.isnan(v)
    except TypeError:
        return False


def _execute_insert(catalog, ins: dict, device=None):
    import numpy as _np
    import pandas as pd

    t = catalog.table(ins["table"], ins["namespace"])
    names = ins["columns"] or t.schema.names()
    if "rows" in ins:
        rows = ins["rows"]
        for r in rows:
            if len(r) != len(names):
                raise SqlError(f"INSERT row has {len(r)} values, expected {len(names)}")
        df = pd.DataFrame(rows, columns=names)
    else:
        df = _execute_select(catalog, ins["select"], device=device)
        if len(df.columns) != len(names):
            raise SqlError("INSERT SELECT column count mismatch")
        df.columns = names
    # cast to schema dtypes; decimal literals are LOGICAL values and
    # scale to the unscaled int64 backing
    for f in t.schema:
        if f.name not in df.columns or not f.is_fixed_width:
            continue
        if f.dtype.startswith("decimal"):
            import decimal as _dec

            from .io.schema import decimal_params

            _, sc = decimal_params(f.dtype)
            vals = [
                None if _is_null_cell(v)
                else int(_dec.Decimal(str(v)).scaleb(sc).to_integral_value())
                for v in df[f.name]
            ]
            # nullable backing so NULL decimals survive
            df[f.name] = (pd.array(vals, dtype="Int64")
                          if any(v is None for v in vals) else vals)
        else:
            from .io.batch import np_dtype_for

            npdt = np_dtype_for(f.dtype)
            if df[f.name].isna().any():
                nullable = {"i": "Int64", "u": "UInt64", "b": "boolean"}.get(
                    _np.dtype(npdt).kind)
                if nullable and _np.dtype(npdt).itemsize <= 8:
                    nullable = {"int8": "Int8", "int16": "Int16", "int32": "Int32",
                                "int64": "Int64", "bool": "boolean",
                                "uint8": "UInt8"}.get(_np.dtype(npdt).name, nullable)
                df[f.name] = (df[f.name].astype(nullable) if nullable
                              else df[f.name].astype(npdt))
            else:
                df[f.name] = df[f.name].astype(npdt)
    t.write(df, device=device)
    return pd.DataFrame({"rows_inserted": [len(df)]})


def _pd_eval(expr: Expr, df, col):
    """Evaluate a filter Expr against a pandas DataFrame (join residuals
    — single-table queries push the same Expr into the scan instead)."""
    import pandas as pd

    if isinstance(expr, And):
        return _pd_eval(expr.left, df, col) & _pd_eval(expr.right, df, col)
    if isinstance(expr, Or):
        return _pd_eval(expr.left, df, col) | _pd_eval(expr.right, df, col)
    if isinstance(expr, Not):
        return ~_pd_eval(expr.inner, df, col)
    if isinstance(expr, IsNull):
        m = df[col(expr.col)].isna()
        return ~m if expr.negate else m
    if isinstance(expr, Cmp):
        series = df[col(expr.col)]
        v = expr.value
        op = expr.op
        if op == "eq":
            return series == v
        if op == "noteq":
            return series != v
        if op == "gt":
            return series > v
        if op == "gteq":
            return series >= v
        if op == "lt":
            return series < v
        if op == "lteq":
            return series <= v
        raise SqlError(f"unsupported join filter op {op}")
    raise SqlError(f"unsupported filter node {type(expr).__name__} in join")


def _execute_join_select(catalog, q: Query, device=None):
    """N-way equi-join (reference: lakesoul-datafusion delegates joins to
    DataFusion; here hash joins folded left-to-right over MOR scans —
    tensor engine by default, pandas as the LAKESOUL_SQL_PANDAS oracle).

    Column naming in the joined frame: the first table to contribute a
    name keeps it bare; later tables' clashing names are prefixed with
    their qualifier ("<alias_or_table>.<col>")."""
    import pandas as pd

    specs = [(q.namespace, q.table, q.alias or q.table, None)] + [
        (j.namespace, j.table, j.alias or j.table, j) for j in q.joins]
    tables = [catalog.table(name, ns) for ns, name, _, _ in specs]
    names_per = [set(t.schema.names()) for t in tables]
    quals = [{alias, name} for (_, name, alias, _) in specs]

    def side_of(name: str):
        """-> (table index, bare column)."""
        if "." in name:
            qual, col = name.split(".", 1)
            for ti in range(len(specs)):
                if qual in quals[ti] and col in names_per[ti]:
                    return ti, col
            raise SqlError(f"cannot resolve {name!r}")
        hits = [ti for ti in range(len(specs)) if name in names_per[ti]]
        if not hits:
            raise SqlError(f"unknown column {name!r}")
        if len(hits) > 1:
            raise SqlError(f"ambiguous column {name!r}: qualify it")
        return hits[0], name

    # referenced columns per table
    refs = []
    star = any(it.kind == "star" for it in q.items)
    out_aliases = {it.out_name for it in q.items} | {
        it.alias for it in q.items if it.alias}
    for it in q.items:
        if it.kind in ("col", "agg") and it.name:
            refs.append(it.name)
        if it.expr is not None:
            refs += list(it.expr.columns())
    refs += q.group_by + [n for n, _ in q.order_by if n not in out_aliases]
    if q.where is not None:
        refs += list(q.where.columns())
    for j in q.joins:
        for a_col, b_col in j.on:
            refs += [a_col, b_col]
    need = [set() for _ in specs]
    for r in refs:
        ti, col = side_of(r)
        need[ti].add(col)
    if star:
        need = [set(n) for n in names_per]

    # frame column mapping: first-come keeps the bare name
    frame_name = {}   # (ti, col) -> column name in the joined frame
    taken = set()
    for ti in range(len(specs)):
        for col in sorted(need[ti]):
            if col in taken:
                frame_name[(ti, col)] = f"{specs[ti][2]}.{col}"
            else:
                frame_name[(ti, col)] = col
                taken.add(col)

    def df_col(name: str) -> str:
        ti, col = side_of(name)
        return frame_name[(ti, col)]

    scans = [
        tables[ti].scan(columns=sorted(need[ti]) or None,
                        version=q.version if ti == 0 else None,
                        device=device)
        for ti in range(len(specs))
    ]

    def resolve_on(j: JoinSpec, new_ti: int):
        """ON pairs -> (acc-frame key names, new-table bare key names)."""
        acc_keys, new_keys = [], []
        for a_col, b_col in j.on:
            sa, ca = side_of(a_col)
            sb, cb = side_of(b_col)
            if sa == new_ti and sb != new_ti:
                sa, ca, sb, cb = sb, cb, sa, ca
            if sb != new_ti or sa == new_ti:
                raise SqlError(
                    "JOIN ON must link the new table to an earlier one")
            acc_keys.append(frame_name[(sa, ca)])
            new_keys.append(cb)
        return acc_keys, new_keys

    if _use_pandas_exec():
        acc = scans[0].to_arrow().to_pandas()
        acc.columns = [frame_name[(0, c)] for c in acc.columns]
        for ti in range(1, len(specs)):
            rdf = scans[ti].to_arrow().to_pandas()
            rdf.columns = [frame_name[(ti, c)] for c in rdf.columns]
            acc_keys, new_keys = resolve_on(specs[ti][3], ti)
            right_on = [frame_name[(ti, c)] for c in new_keys]
            acc = acc.merge(rdf, how=specs[ti][3].kind,
                            left_on=acc_keys, right_on=right_on)
        if q.where is not None:
            acc = acc[_pd_eval(q.where, acc, df_col)].reset_index(drop=True)
        return _project_and_finish(q, acc, df_col, all_cols=list(acc.columns))

    # tensor engine path (vectorized hash join, query/engine.py)
    import torch as _torch

    from .io.batch import Batch as _B
    from .io.schema import Field as _F
    from .io.schema import Schema as _S
    from .query.engine import join_batches

    def renamed_batch(ti):
        b = scans[ti].to_batch()
        fields, cols = [], {}
        for f in b.schema:
            nn = frame_name[(ti, f.name)]
            fields.append(_F(nn, f.dtype, f.nullable))
            cols[nn] = b.columns[f.name]
        return _B(_S(fields), cols)

    acc = renamed_batch(0)
    for ti in range(1, len(specs)):
        rb = renamed_batch(ti)
        acc_keys, new_keys = resolve_on(specs[ti][3], ti)
        right_on = [frame_name[(ti, c)] for c in new_keys]
        acc = join_batches(acc, rb, acc_keys, right_on, specs[ti][3].kind)
    if q.where is not None:
        view = _expr_view_batch(acc, q.where, df_col)
        mask = q.where.evaluate(view)
        acc = acc.take(_torch.nonzero(mask, as_tuple=False).flatten())
    return _project_and_finish_tensor(
        q, acc, df_col, all_cols=list(acc.schema.names()))


def _eval_scalar_tensor(e: Scalar, batch, col):
    """Scalar expression -> (values (n,) float64 tensor, validity or
    None). Decimal columns are scaled to their logical value; nulls
    propagate through arithmetic."""
    import torch as _torch

    n = batch.num_rows
    if e.kind == "num":
        dev = None
        for c in batch.columns.values():
            t = c.offsets if c.is_string else c.data
            if t is not None:
                dev = t.device
                break
        return (_torch.full((n,), float(e.value), dtype=_torch.float64,
                            device=dev or "cpu"), None)
    if e.kind == "col":
        src = col(e.name)
        c = batch.columns[src]
        f = batch.schema.field(src)
        if c.is_string:
            raise SqlError(f"arithmetic on string column {e.name!r}")
        v = c.data.to(_torch.float64)
        if f.dtype.startswith("decimal"):
            from .io.schema import decimal_params

            _, sc = decimal_params(f.dtype)
            v = v / (10 ** sc)
        return v, c.validity
    l, lv = _eval_scalar_tensor(e.left, batch, col)
    r, rv = _eval_scalar_tensor(e.right, batch, col)
    if e.op == "+":
        out = l + r
    elif e.op == "-":
        out = l - r
    elif e.op == "*":
        out = l * r
    elif e.op == "/":
        out = l / r
    else:
        raise SqlError(f"unknown operator {e.op!r}")
    if lv is None and rv is None:
        return out, None
    import torch as _torch

    m = _torch.ones(len(out), dtype=_torch.bool, device=out.device)
    if lv is not None:
        m &= lv.to(_torch.bool)
    if rv is not None:
        m &= rv.to(_torch.bool)
    return out, m.to(_torch.uint8)


def _eval_scalar_pd(e: Scalar, df, col):
    if e.kind == "num":
        return e.value
    if e.kind == "col":
        import pandas as _pd

        s = df[col(e.name)]
        if len(s) and isinstance(s.iloc[0], __import__("decimal").Decimal):
            s = s.astype(float)
        return s
    l = _eval_scalar_pd(e.left, df, col)
    r = _eval_scalar_pd(e.right, df, col)
    if e.op == "+":
        return l + r
    if e.op == "-":
        return l - r
    if e.op == "*":
        return l * r
    return l / r


def _use_pandas_exec() -> bool:
    """Escape hatch: LAKESOUL_SQL_PANDAS=1 selects the legacy pandas
    execution (kept as the cross-check oracle; the default is the tensor
    engine, lakesoul_amd/query/engine.py)."""
    import os

    return os.environ.get("LAKESOUL_SQL_PANDAS", "0") == "1"


def _expr_view_batch(batch, expr: Expr, col):
    """A Batch whose column names match the expression's references
    (qualified names resolved through ``col``)."""
    from .io.batch import Batch as _B
    from .io.schema import Field as _F
    from .io.schema import Schema as _S

    fields, cols = [], {}
    for name in sorted(expr.columns()):
        src = col(name)
        c = batch.columns[src]
        f = batch.schema.field(src)
        fields.append(_F(name, f.dtype, f.nullable))
        cols[name] = c
    return _B(_S(fields), cols)


def _project_and_finish_tensor(q: Query, batch, col, all_cols):
    """Tensor-engine SELECT tail: aggregation / projection / DISTINCT /
    HAVING / ORDER BY / LIMIT over a Batch; pandas materializes only the
    final (small) result."""
    import torch as _torch

    from .io.batch import Batch as _B
    from .io.schema import Field as _F
    from .io.schema import Schema as _S
    from .query.engine import distinct_indices, groupby_agg, sort_indices

    # materialize computed scalar expressions as extra float64 columns so
    # the aggregation/projection below sees plain columns
    expr_names = {}
    expr_i = 0
    for it in q.items:
        if it.expr is not None:
            key = id(it)
            name = f"__sx{expr_i}"
            expr_i += 1
            from .utils import timing as _tm0

            with _tm0.phase("sql_expr_eval", sync_gpu=True):
                vals, validity = _eval_scalar_tensor(it.expr, batch, col)
            from .io.batch import Column as _C

            nf = list(batch.schema.fields) + [_F(name, "float64", True)]
            nc = dict(batch.columns)
            nc[name] = _C("float64", data=vals, validity=validity)
            batch = _B(_S(nf), nc)
            expr_names[key] = name

    def _arg(it):
        if it.expr is not None:
            return expr_names[id(it)]
        return col(it.name) if it.name else None

    has_agg = any(it.kind == "agg" for it in q.items)
    if has_agg or q.group_by:
        gcols = [col(g) for g in q.group_by]
        aggs = []
        out_order = []
        for it in q.items:
            if it.kind == "col" and it.expr is None:
                if it.name not in q.group_by:
                    raise SqlError(f"column {it.name!r} must appear in GROUP BY")
                out_order.append(("group", col(it.name),
                                  it.alias or it.name.split(".")[-1]))
                continue
            if it.kind == "star":
                raise SqlError("SELECT * with aggregates is not valid")
            if it.kind == "col":
                raise SqlError("computed columns must be aggregated "
                               "when GROUP BY is present")
            aggs.append((it.fn, _arg(it), it.out_name, it.distinct))
            out_order.append(("agg", it.out_name, it.out_name))
        from .utils import timing as _tm1

        with _tm1.phase("sql_groupby_agg", sync_gpu=True):
            res = groupby_agg(batch, gcols, aggs)
        fields, cols = [], {}
        for kind, src, outn in out_order:
            f = res.schema.field(src)
            fields.append(_F(outn, f.dtype, f.nullable))
            cols[outn] = res.columns[src]
        out = _B(_S(fields), cols)
    else:
        fields, cols = [], {}
        for it in q.items:
            if it.kind == "star":
                for c in all_cols:
                    if c not in cols:
                        fields.append(_F(c, batch.schema.field(c).dtype,
                                         batch.schema.field(c).nullable))
                        cols[c] = batch.columns[c]
            elif it.expr is not None:
                outn = it.out_name
                if outn not in cols:
                    src = expr_names[id(it)]
                    fields.append(_F(outn, "float64", True))
                    cols[outn] = batch.columns[src]
            else:
                src = col(it.name)
                outn = it.alias or it.name.split(".")[-1]
                if outn not in cols:
                    f = batch.schema.field(src)
                    fields.append(_F(outn, f.dtype, f.nullable))
                    cols[outn] = batch.columns[src]
        out = _B(_S(fields), cols)
        if q.distinct:
            out = out.take(distinct_indices(out, [f.name for f in out.schema]))

    if q.having is not None:
        view = _expr_view_batch(out, q.having, lambda n: (
            n if n in out.schema.names() else n.split(".")[-1]))
        mask = q.having.evaluate(view)
        out = out.take(_torch.nonzero(mask, as_tuple=False).flatten())
    if q.order_by:
        by = []
        for n, d in q.order_by:
            if n in out.schema.names():
                by.append((n, not d))
            else:
                cn = col(n)
                if cn in out.schema.names():
                    by.append((cn, not d))
                else:
                    base = n.split(".")[-1]
                    if base in out.schema.names():
                        by.append((base, not d))
                    else:
                        raise SqlError(f"ORDER BY references unknown column {n!r}")
        out = out.take(sort_indices(out, by))
    a = q.offset or 0
    b = out.num_rows if q.limit is None else min(out.num_rows, a + q.limit)
    if a or b < out.num_rows:
        out = out.slice(a, max(a, b))
    return out.to_arrow().to_pandas()


def _project_and_finish(q: Query, df, col, all_cols):
    """Shared SELECT tail: aggregation / projection / DISTINCT / ORDER BY
    / LIMIT over a materialized DataFrame. ``col`` maps a referenced name
    to the DataFrame column holding it."""
    import pandas as pd

    has_agg = any(it.kind == "agg" for it in q.items)
    if has_agg or q.group_by:
        def agg_row(sub: pd.DataFrame):
            row = {}
            for it in q.items:
                if it.kind == "col":
                    if it.name not in q.group_by:
                        raise SqlError(
                            f"column {it.name!r} must appear in GROUP BY")
                    continue
                if it.kind == "star":
                    raise SqlError("SELECT * with aggregates is not valid")
                if it.expr is not None:
                    s = _eval_scalar_pd(it.expr, sub, col)
                else:
                    s = sub[col(it.name)] if it.name else None
                if it.fn == "count":
                    if it.distinct and s is not None:
                        row[it.out_name] = int(s.dropna().nunique())
                    else:
                        row[it.out_name] = len(sub) if s is None else int(s.notna().sum())
                elif it.fn == "sum":
                    row[it.out_name] = s.sum()
                elif it.fn == "min":
                    row[it.out_name] = s.min()
                elif it.fn == "max":
                    row[it.out_name] = s.max()
                elif it.fn == "avg":
                    row[it.out_name] = float(s.mean())
            return row

        def agg_series(sub: pd.DataFrame):
            return pd.Series(agg_row(sub))

        if q.group_by:
            gcols = [col(g) for g in q.group_by]
            out = (df.groupby(gcols, as_index=False, sort=False)
                     .apply(agg_series, include_groups=False)
                   if pd.__version__ >= "2.2"
                   else df.groupby(gcols, as_index=False).apply(agg_series))
            cols = []
            for it in q.items:
                if it.kind == "col":
                    dfc = col(it.name)
                    outn = it.alias or it.name.split(".")[-1]
                    if dfc != outn:
                        out = out.rename(columns={dfc: outn})
                    cols.append(outn)
                else:
                    cols.append(it.out_name)
            out = out[cols]
        else:
            # dict-of-columns keeps count() integral next to float aggs
            out = pd.DataFrame({k: [v] for k, v in agg_row(df).items()})
    else:
        cols, ren = [], {}
        computed = {}
        for it in q.items:
            if it.kind == "star":
                for c in all_cols:
                    if c not in cols:
                        cols.append(c)
            elif it.expr is not None:
                computed[it.out_name] = _eval_scalar_pd(it.expr, df, col)
            else:
                dfc = col(it.name)
                cols.append(dfc)
                outn = it.alias or it.name.split(".")[-1]
                if dfc != outn:
                    ren[dfc] = outn
        out = df[cols].rename(columns=ren)
        for name, s in computed.items():
            out[name] = s
        if q.distinct:
            out = out.drop_duplicates().reset_index(drop=True)

    if q.having is not None:
        # HAVING references output columns (aggregate aliases / group cols)
        def having_col(name):
            if name in out.columns:
                return name
            base = name.split(".")[-1]
            if base in out.columns:
                return base
            raise SqlError(f"HAVING references unknown output column {name!r}")

        out = out[_pd_eval(q.having, out, having_col)].reset_index(drop=True)
    if q.order_by:
        names = []
        for n, _ in q.order_by:
            # ORDER BY may reference an output alias or a source column
            names.append(n if n in out.columns
                         else (col(n) if col(n) in out.columns else n))
        asc = [not d for _, d in q.order_by]
        out = out.sort_values(names, ascending=asc).reset_index(drop=True)
    if q.offset:
        out = out.iloc[q.offset:].reset_index(drop=True)
    if q.limit is not None:
        out = out.head(q.limit).reset_index(drop=True)
    return out.reset_index(drop=True)


def _strip_quals(q: Query, valid_quals, passthrough=()) -> None:
    """Single-table queries: rewrite `t.col`/`alias.col` to `col`.
    Dotted names whose first part is a STRUCT column (``st.a``) pass
    through untouched — they resolve as member access downstream."""
    def strip(name: str) -> str:
        if "." in name:
            qual, col = name.split(".", 1)
            if qual in valid_quals:
                return col
            if qual in passthrough:
                return name
            raise SqlError(f"unknown table qualifier {qual!r}")
        return name

    def strip_scalar(e):
        if e is None:
            return
        if e.kind == "col":
            e.name = strip(e.name)
        elif e.kind == "bin":
            strip_scalar(e.left)
            strip_scalar(e.right)

    for it in q.items:
        if it.kind in ("col", "agg") and it.name:
            it.name = strip(it.name)
        strip_scalar(it.expr)
    q.group_by = [strip(g) for g in q.group_by]
    q.order_by = [(strip(n) if "." in n else n, d) for n, d in q.order_by]

    def walk(e):
        if e is None:
            return
        if isinstance(e, (And, Or)):
            walk(e.left)
            walk(e.right)
        elif isinstance(e, Not):
            walk(e.inner)
        elif isinstance(e, (Cmp, IsNull)):
            e.col = strip(e.col)

    walk(q.where)


def _explain_select(catalog, q: Query, device=None):
    """EXPLAIN: the physical scan plan — per-unit file counts after
    partition/stats/bucket pruning, pushdown summary (the reference shows
    DataFusion's plan; this shows ours)."""
    import pandas as pd

    q.where = _resolve_subqueries(catalog, q.where, device)
    rows = []
    for tbl_name, ns, flt in ((q.table, q.namespace, q.where),) + tuple(
        (j.table, j.namespace, None) for j in q.joins
    ):
        t = catalog.table(tbl_name, ns)
        scan = t.scan(filters=flt, version=q.version,
                      timestamp_ms=q.timestamp_ms, device=device)
        units = scan.plan()
        total_files = sum(len(u.files) for u in units)
        rows.append({
            "table": f"{ns}.{tbl_name}",
            "scan_units": len(units),
            "files": total_files,
            "hash_buckets": t.hash_bucket_num,
            "pushdown": str(flt) if flt is not None else "",
            "device": scan.device,
        })
    return pd.DataFrame(rows)


def _resolve_subqueries(catalog, expr, device):
    """Execute non-correlated subqueries and substitute literal sets /
    scalars (the reference delegates this to DataFusion's planner)."""
    if expr is None:
        return None
    if isinstance(expr, (And, Or)):
        expr.left = _resolve_subqueries(catalog, expr.left, device)
        expr.right = _resolve_subqueries(catalog, expr.right, device)
        return expr
    if isinstance(expr, Not):
        expr.inner = _resolve_subqueries(catalog, expr.inner, device)
        return expr
    if isinstance(expr, SubqueryIn):
        sub = _execute_select(catalog, expr.query, device=device)
        if len(sub.columns) != 1:
            raise SqlError("IN subquery must return one column")
        vals = [v for v in sub.iloc[:, 0].tolist() if v is not None]
        if not vals:
            return Literal(False)
        e: Expr = Cmp(expr.col, "eq", vals[0])
        for x in vals[1:]:
            e = Or(e, Cmp(expr.col, "eq", x))
        return e
    if isinstance(expr, SubqueryCmp):
        sub = _execute_select(catalog, expr.query, device=device)
        if len(sub.columns) != 1 or len(sub) != 1:
            raise SqlError("scalar subquery must return exactly one value")
        return Cmp(expr.col, expr.op, sub.iloc[0, 0])
    return expr


def _execute_select(catalog, q: Query, device=None):
    import pandas as pd

    q.where = _resolve_subqueries(catalog, q.where, device)
    if q.joins:
        return _execute_join_select(catalog, q, device=device)
    t = catalog.table(q.table, q.namespace)
    _strip_quals(q, {q.table, q.alias} - {""},
                 passthrough={f.name for f in t.schema
                              if f.dtype.startswith("struct<")})
    schema_cols = t.schema.names()

    # columns actually needed from storage
    need = set(q.group_by)
    star = any(it.kind == "star" for it in q.items)
    has_agg = any(it.kind == "agg" for it in q.items)
    for it in q.items:
        if it.kind in ("col", "agg") and it.name:
            need.add(it.name)
        if it.expr is not None:
            need |= it.expr.columns()
    for name, _ in q.order_by:
        if name in schema_cols:
            need.add(name)
    if star:
        need = set(schema_cols)
    # struct member access: SELECT st.a resolves to the struct column's
    # child (Spark-style dotted access; the scan reads the struct and the
    # member materializes as a flat column named "st.a")
    from .io.schema import struct_members as _smem

    member_refs = set()
    for c in list(need):
        if c in schema_cols:
            continue
        base, _, mem = c.partition(".")
        sm = (_smem(t.schema.field(base).dtype)
              if base in schema_cols else None)
        if sm is not None and any(mn == mem for mn, _ in sm):
            member_refs.add(c)
            need.discard(c)
            need.add(base)
            continue
        raise SqlError(f"unknown column {c!r} in {q.namespace}.{q.table}")

    # count(*) with no WHERE/grouping: metadata-free count-only scan
    # (reference: EmptyScanCountExec, physical_plan/empty_schema.rs:192)
    if (len(q.items) == 1 and q.items[0].kind == "agg"
            and q.items[0].fn == "count" and not q.items[0].name
            and q.where is None and not q.group_by and not q.distinct):
        n = t.scan(version=q.version, timestamp_ms=q.timestamp_ms,
                   device=device).count()
        return pd.DataFrame({q.items[0].out_name: [n]})

    scan = t.scan(columns=sorted(need) or None, filters=q.where,
                  version=q.version, timestamp_ms=q.timestamp_ms,
                  device=device)
    if _use_pandas_exec():
        df = scan.to_arrow().to_pandas()
        for r in member_refs:
            base, _, mem = r.partition(".")
            df[r] = df[base].apply(
                lambda d: None if d is None or (not isinstance(d, dict)
                                                and pd.isna(d)) else d[mem])
        return _project_and_finish(q, df, lambda n: n, all_cols=schema_cols)
    from .utils import timing as _tm

    with _tm.phase("sql_scan_to_batch", sync_gpu=True):
        batch = scan.to_batch()
    if member_refs:
        batch = _materialize_struct_members(batch, member_refs)
    return _project_and_finish_tensor(q, batch, lambda n: n,
                                      all_cols=[c for c in schema_cols
                                                if c in batch.schema.names()])


def _materialize_struct_members(batch, refs):
    """Lift struct children referenced as ``base.member`` into flat
    columns of the batch (the struct's validity governs nullability)."""
    from .io.batch import Batch as _B, Column as _C
    from .io.schema import Field as _F, Schema as _S

    fields = list(batch.schema.fields)
    cols = dict(batch.columns)
    for r in sorted(refs):
        base, _, mem = r.partition(".")
        c = cols[base]
        ch = c.children[mem]
        cols[r] = _C(ch.dtype, data=ch.data, offsets=ch.offsets,
                     bytes_=ch.bytes_, elem_offsets=ch.elem_offsets,
                     validity=c.validity)
        fields.append(_F(r, ch.dtype, True))
    return _B(_S(fields), cols)


def repl(catalog, device=None, input_fn=input, print_fn=print):
    """Interactive console loop (reference: lakesoul-console/src/main.rs)."""
    print_fn("lakesoul_amd SQL console — \\q to quit")
    while True:
        try:
            line = input_fn("lakesoul> ")
        except (EOFError, KeyboardInterrupt):
            break
        line = line.strip()
        if not line:
            continue
        if line in ("\\q", "quit", "exit"):
            break
        try:
            df = execute_sql(catalog, line, device=device)
            try:
                from tabulate import tabulate
                print_fn(tabulate(df, headers="keys", tablefmt="psql", showindex=False))
            except ImportError:
                print_fn(df.to_string(index=False))
            print_fn(f"({len(df)} rows)")
        except Exception as e:  # console: report, keep looping
            print_fn(f"error: {e}")
