"""SQL tokenizer + recursive-descent parser producing a small AST.

Replaces the reference's Calcite JVM frontend (BodoSQL/calcite_sql, SURVEY
§2.8: "the Calcite JVM can be replaced by reusing the same logical-plan
builder") with a native parser for the analytic subset the engine executes:
SELECT / FROM / JOIN / WHERE / GROUP BY / HAVING / ORDER BY / LIMIT,
standard expressions, CASE, IN, BETWEEN, LIKE, EXTRACT, CAST.
"""

from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Any, List, Optional, Tuple

KEYWORDS = {
    "select", "distinct", "from", "where", "group", "by", "having", "order",
    "limit", "as", "and", "or", "not", "in", "between", "like", "is", "null",
    "case", "when", "then", "else", "end", "cast", "extract", "join", "inner",
    "left", "right", "full", "outer", "cross", "on", "asc", "desc", "date",
    "interval", "union", "all", "exists", "true", "false", "substring", "for",
    "intersect", "except", "over", "partition",
}

_TOKEN_RE = re.compile(r"""
    (?P<ws>\s+)
  | (?P<num>\d+\.\d*|\.\d+|\d+)
  | (?P<str>'(?:[^']|'')*')
  | (?P<id>[A-Za-z_][A-Za-z_0-9]*)
  | (?P<op><>|<=|>=|!=|\|\||[-+*/%(),.<>=])
""", re.VERBOSE)


#: identifiers that start a clause and therefore can never be an implicit
#: table alias (dialect words kept out of KEYWORDS for expression use)
_NON_ALIAS_IDS = {"qualify", "offset", "rows", "window", "fetch"}


@dataclass
class Tok:
    kind: str  # num, str, id, kw, op
    value: str


def tokenize(sql: str) -> List[Tok]:
    out = []
    pos = 0
    sql = re.sub(r"--[^\n]*", " ", sql)
    while pos < len(sql):
        m = _TOKEN_RE.match(sql, pos)
        if not m:
            raise SyntaxError(f"bad SQL at: {sql[pos:pos + 30]!r}")
        pos = m.end()
        if m.lastgroup == "ws":
            continue
        v = m.group()
        if m.lastgroup == "id":
            low = v.lower()
            out.append(Tok("kw", low) if low in KEYWORDS else Tok("id", v))
        elif m.lastgroup == "str":
            out.append(Tok("str", v[1:-1].replace("''", "'")))
        else:
            out.append(Tok(m.lastgroup, v))
    return out


# ---------------------------------------------------------------- AST
@dataclass
class Col:
    table: Optional[str]
    name: str


@dataclass
class Lit:
    value: Any
    kind: str = "auto"  # num/str/date/bool/null


@dataclass
class Bin:
    op: str
    left: Any
    right: Any


@dataclass
class Un:
    op: str
    operand: Any


@dataclass
class Func:
    name: str
    args: List[Any]
    distinct: bool = False
    star: bool = False


@dataclass
class CaseE:
    whens: List[Tuple[Any, Any]]
    els: Optional[Any]


@dataclass
class InE:
    operand: Any
    values: List[Any]
    negated: bool = False


@dataclass
class BetweenE:
    operand: Any
    lo: Any
    hi: Any
    negated: bool = False


@dataclass
class LikeE:
    operand: Any
    pattern: str
    negated: bool = False
    ci: bool = False  # ILIKE


@dataclass
class IsNullE:
    operand: Any
    negated: bool = False


@dataclass
class CastE:
    operand: Any
    to: str
    safe: bool = False  # TRY_CAST: NULL on conversion failure


@dataclass
class ExtractE:
    fld: str
    operand: Any


@dataclass
class SubqueryE:
    """Scalar subquery (SELECT agg ...)."""

    query: "Query"


@dataclass
class InSubquery:
    operand: Any
    query: "Query"
    negated: bool = False


@dataclass
class ExistsE:
    query: "Query"
    negated: bool = False


@dataclass
class WindowE:
    """func(args) OVER (PARTITION BY ... ORDER BY ... [ROWS BETWEEN ...]).
    frame_preceding: None = no frame clause; -1 = UNBOUNDED PRECEDING;
    n >= 0 = n PRECEDING (frame end is always CURRENT ROW)."""

    func: str
    args: List[Any]
    partition_by: List[Any]
    order_by: List[Tuple[Any, bool]]
    star: bool = False
    frame_preceding: Optional[int] = None
    distinct: bool = False


@dataclass
class SetOpQ:
    """UNION / INTERSECT / EXCEPT of two query expressions."""

    op: str
    all: bool
    left: Any
    right: Any
    ctes: Optional[List[Tuple[str, Any]]] = None  # WITH name AS (...)


@dataclass
class TableRef:
    name: str
    alias: Optional[str]
    subquery: Optional["Query"] = None
    flatten: Any = None  # LATERAL FLATTEN(input => expr)


@dataclass
class JoinClause:
    kind: str  # inner/left/right/full/cross
    table: TableRef
    on: Optional[Any]


@dataclass
class SelectItem:
    expr: Any
    alias: Optional[str]
    star: bool = False
    star_table: Optional[str] = None  # qualified star: t.*


@dataclass
class Query:
    items: List[SelectItem]
    distinct: bool
    table: Optional[TableRef]
    joins: List[JoinClause]
    where: Optional[Any]
    group_by: List[Any]
    having: Optional[Any]
    order_by: List[Tuple[Any, bool]]
    limit: Optional[int]
    grouping_sets: Optional[List[List[Any]]] = None
    qualify: Optional[Any] = None  # Snowflake-dialect window filter
    ctes: Optional[List[Tuple[str, Any]]] = None  # WITH name AS (...)


class Parser:
    def __init__(self, sql: str):
        self.toks = tokenize(sql)
        self.i = 0

    # ------------------------------------------------------------- utils
    def peek(self) -> Optional[Tok]:
        return self.toks[self.i] if self.i < len(self.toks) else None

    def next(self) -> Tok:
        t = self.peek()
        if t is None:
            raise SyntaxError("unexpected end of SQL")
        self.i += 1
        return t

    def accept_kw(self, *kws) -> Optional[str]:
        t = self.peek()
        if t and t.kind == "kw" and t.value in kws:
            self.i += 1
            return t.value
        return None

    def expect_kw(self, kw):
        if not self.accept_kw(kw):
            raise SyntaxError(f"expected {kw.upper()} at {self.peek()}")

    def accept_op(self, op) -> bool:
        t = self.peek()
        if t and t.kind == "op" and t.value == op:
            self.i += 1
            return True
        return False

    def expect_op(self, op):
        if not self.accept_op(op):
            raise SyntaxError(f"expected {op!r} at {self.peek()}")

    # ------------------------------------------------------------- query
    def parse(self):
        ctes = None
        t = self.peek()
        if t and t.kind == "id" and t.value.lower() == "with":
            # WITH name AS (query) [, name2 AS (query2)] ... SELECT ...
            self.i += 1
            ctes = []
            while True:
                name = self.next().value.lower()
                self.expect_kw("as")
                self.expect_op("(")
                sub = self.parse_query_expr()
                self.expect_op(")")
                ctes.append((name, sub))
                if not self.accept_op(","):
                    break
        q = self.parse_query_expr()
        if self.peek() is not None:
            raise SyntaxError(f"trailing tokens at {self.peek()}")
        q.ctes = ctes
        return q

    def parse_query_expr(self):
        q = self.parse_select()
        while True:
            op = None
            for kw in ("union", "intersect", "except"):
                if self.accept_kw(kw):
                    op = kw
                    break
            if op is None:
                return q
            all_ = bool(self.accept_kw("all"))
            self.accept_kw("distinct")
            r = self.parse_select()
            q = SetOpQ(op, all_, q, r)

    def parse_select(self) -> Query:
        self.expect_kw("select")
        distinct = bool(self.accept_kw("distinct"))
        top_n = None
        tt = self.peek()
        if tt and tt.kind == "id" and tt.value.lower() == "top":
            nt = self.toks[self.i + 1] if self.i + 1 < len(self.toks) else None
            if nt and nt.kind == "num":
                self.i += 2
                top_n = int(nt.value)
        items = [self.parse_select_item()]
        while self.accept_op(","):
            items.append(self.parse_select_item())
        table, joins = None, []
        if self.accept_kw("from"):
            table = self.parse_table_ref()
            while True:
                t = self.peek()
                if t is None:
                    break
                if t.kind == "op" and t.value == ",":
                    self.i += 1
                    nt = self.peek()
                    if nt and nt.kind == "id" and \
                            nt.value.lower() == "lateral":
                        joins.append(JoinClause(
                            "lateral", self.parse_lateral_flatten(), None))
                        continue
                    joins.append(JoinClause("cross", self.parse_table_ref(), None))
                    continue
                kind = None
                if self.accept_kw("inner"):
                    kind = "inner"
                elif self.accept_kw("left"):
                    self.accept_kw("outer")
                    kind = "left"
                elif self.accept_kw("right"):
                    self.accept_kw("outer")
                    kind = "right"
                elif self.accept_kw("full"):
                    self.accept_kw("outer")
                    kind = "outer"
                elif self.accept_kw("cross"):
                    kind = "cross"
                if kind is None:
                    if self.peek() and self.peek().kind == "kw" \
                            and self.peek().value == "join":
                        kind = "inner"
                    else:
                        break
                self.expect_kw("join")
                tr = self.parse_table_ref()
                on = None
                if kind != "cross" and self.accept_kw("on"):
                    on = self.parse_expr()
                joins.append(JoinClause(kind, tr, on))
        where = self.parse_expr() if self.accept_kw("where") else None
        group_by: List[Any] = []
        grouping_sets = None
        if self.accept_kw("group"):
            self.expect_kw("by")
            if self.accept_kw("all"):
                group_by = "ALL"  # resolved by the planner to non-agg items
                t = None
            else:
                t = self.peek()
            word = t.value.lower() if t is not None and t.kind == "id" else ""
            nxt = self.toks[self.i + 1] if self.i + 1 < len(self.toks) else None
            if word in ("rollup", "cube") and nxt and nxt.kind == "op" \
                    and nxt.value == "(":
                self.i += 1
                self.expect_op("(")
                group_by.append(self.parse_expr())
                while self.accept_op(","):
                    group_by.append(self.parse_expr())
                self.expect_op(")")
                if word == "rollup":
                    grouping_sets = [group_by[:i]
                                     for i in range(len(group_by), -1, -1)]
                else:  # cube: the full powerset
                    import itertools as _it

                    grouping_sets = [list(c) for r in
                                     range(len(group_by), -1, -1)
                                     for c in _it.combinations(group_by, r)]
            elif word == "grouping" and nxt and nxt.kind == "id" \
                    and nxt.value.lower() == "sets":
                self.i += 2
                self.expect_op("(")
                grouping_sets = []
                while True:
                    self.expect_op("(")
                    gset = []
                    if not self.accept_op(")"):
                        gset.append(self.parse_expr())
                        while self.accept_op(","):
                            gset.append(self.parse_expr())
                        self.expect_op(")")
                    grouping_sets.append(gset)
                    if not self.accept_op(","):
                        break
                self.expect_op(")")
                seen = []
                for gset in grouping_sets:
                    for g in gset:
                        if repr(g) not in {repr(x) for x in seen}:
                            seen.append(g)
                group_by = seen
            elif group_by != "ALL":
                group_by.append(self.parse_expr())
                while self.accept_op(","):
                    group_by.append(self.parse_expr())
        having = self.parse_expr() if self.accept_kw("having") else None
        qualify = None
        qt = self.peek()
        if qt and qt.kind == "id" and qt.value.lower() == "qualify":
            self.i += 1
            qualify = self.parse_expr()
        order_by: List[Tuple[Any, bool]] = []
        if self.accept_kw("order"):
            self.expect_kw("by")
            if self.accept_kw("all"):
                order_by = "ALL"
            else:
                pass
            while order_by != "ALL":
                e = self.parse_expr()
                asc = True
                if self.accept_kw("desc"):
                    asc = False
                else:
                    self.accept_kw("asc")
                nt = self.peek()
                if nt and nt.kind == "id" and nt.value.lower() == "nulls":
                    self.i += 1
                    self.next()  # FIRST | LAST (engine default preserved)
                order_by.append((e, asc))
                if not self.accept_op(","):
                    break
        limit = None
        limit_offset = 0
        if self.accept_kw("limit"):
            limit = int(self.next().value)
            nt = self.peek()
            if nt and nt.kind == "id" and nt.value.lower() == "offset":
                self.i += 1
                limit_offset = int(self.next().value)
        if top_n is not None and limit is None:
            limit = top_n
        q = Query(items, distinct, table, joins, where, group_by, having,
                  order_by, limit, grouping_sets, qualify)
        q.limit_offset = limit_offset
        return q

    def parse_lateral_flatten(self) -> TableRef:
        """, LATERAL FLATTEN([input =>] expr) [AS] alias  (Snowflake
        surface; reference: bodo/libs/_lateral.cpp FLATTEN)."""
        self.next()  # lateral
        fn = self.next()
        if fn.kind != "id" or fn.value.lower() != "flatten":
            raise SyntaxError("expected FLATTEN after LATERAL")
        self.expect_op("(")
        nt = self.peek()
        if nt and nt.kind == "id" and nt.value.lower() == "input":
            self.next()
            self.expect_op("=")
            self.expect_op(">")
        e = self.parse_expr()
        self.expect_op(")")
        alias = None
        if self.accept_kw("as"):
            alias = self.next().value
        else:
            nt = self.peek()
            if nt and nt.kind == "id" \
                    and nt.value.lower() not in _NON_ALIAS_IDS:
                alias = self.next().value
        return TableRef("__flatten", alias or "f", flatten=e)

    def parse_table_ref(self) -> TableRef:
        if self.accept_op("("):
            q = self.parse_query_expr()
            self.expect_op(")")
            alias = None
            if self.accept_kw("as"):
                alias = self.next().value
            else:
                nt = self.peek()
                if nt and nt.kind == "id" \
                        and nt.value.lower() not in _NON_ALIAS_IDS:
                    alias = self.next().value
            return TableRef(alias or "__subq", alias, subquery=q)
        t = self.next()
        if t.kind != "id":
            raise SyntaxError(f"expected table name, got {t}")
        alias = None
        nt = self.peek()
        if nt and nt.kind == "id" and nt.value.lower() not in _NON_ALIAS_IDS:
            alias = self.next().value
        elif self.accept_kw("as"):
            alias = self.next().value
        return TableRef(t.value, alias)

    def parse_select_item(self) -> SelectItem:
        if self.accept_op("*"):
            nt = self.peek()
            if nt and nt.kind == "id" and nt.value.lower() == "exclude":
                self.i += 1
                bracket = bool(self.accept_op("("))
                cols = [self.next().value]
                while self.accept_op(","):
                    cols.append(self.next().value)
                if bracket:
                    self.expect_op(")")
                it = SelectItem(None, None, star=True)
                it.exclude = tuple(c.lower() for c in cols)
                return it
            return SelectItem(None, None, star=True)
        t = self.peek()
        nt = self.toks[self.i + 1] if self.i + 1 < len(self.toks) else None
        n3 = self.toks[self.i + 2] if self.i + 2 < len(self.toks) else None
        if t and t.kind == "id" and nt and nt.kind == "op" \
                and nt.value == "." and n3 and n3.kind == "op" \
                and n3.value == "*":
            self.i += 3
            return SelectItem(None, None, star=True, star_table=t.value)
        e = self.parse_expr()
        alias = None
        if self.accept_kw("as"):
            alias = self.next().value
        else:
            nt = self.peek()
            if nt and nt.kind == "id":
                alias = self.next().value
        return SelectItem(e, alias)

    # --------------------------------------------------------- expressions
    def parse_expr(self):
        return self.parse_or()

    def parse_or(self):
        left = self.parse_and()
        while self.accept_kw("or"):
            left = Bin("or", left, self.parse_and())
        return left

    def parse_and(self):
        left = self.parse_not()
        while self.accept_kw("and"):
            left = Bin("and", left, self.parse_not())
        return left

    def parse_not(self):
        if self.accept_kw("not"):
            return Un("not", self.parse_not())
        return self.parse_predicate()

    def parse_predicate(self):
        left = self.parse_add()
        t = self.peek()
        negated = False
        if t and t.kind == "kw" and t.value == "not":
            nxt = self.toks[self.i + 1] if self.i + 1 < len(self.toks) else None
            if nxt and ((nxt.kind == "kw" and nxt.value in
                         ("in", "like", "between"))
                        or (nxt.kind == "id" and nxt.value.lower() in
                            ("ilike", "rlike", "regexp"))):
                self.i += 1
                negated = True
                t = self.peek()
        if t and t.kind == "op" and t.value in ("=", "<>", "!=", "<", "<=", ">", ">="):
            self.i += 1
            opmap = {"=": "eq", "<>": "ne", "!=": "ne", "<": "lt",
                     "<=": "le", ">": "gt", ">=": "ge"}
            return Bin(opmap[t.value], left, self.parse_add())
        if t and t.kind == "kw" and t.value == "in":
            self.i += 1
            self.expect_op("(")
            nt = self.peek()
            if nt and nt.kind == "kw" and nt.value == "select":
                q = self.parse_select()
                self.expect_op(")")
                return InSubquery(left, q, negated)
            vals = [self.parse_expr()]
            while self.accept_op(","):
                vals.append(self.parse_expr())
            self.expect_op(")")
            return InE(left, vals, negated)
        if t and t.kind == "kw" and t.value == "between":
            self.i += 1
            lo = self.parse_add()
            self.expect_kw("and")
            hi = self.parse_add()
            return BetweenE(left, lo, hi, negated)
        if t and t.kind == "kw" and t.value == "like":
            self.i += 1
            nt = self.peek()
            if nt and nt.value.lower() == "any":  # LIKE ANY (p1, p2, ...)
                self.i += 1
                self.expect_op("(")
                pats = [self.next().value]
                while self.accept_op(","):
                    pats.append(self.next().value)
                self.expect_op(")")
                e = LikeE(left, pats[0], False)
                for pt in pats[1:]:
                    e = Bin("or", e, LikeE(left, pt, False))
                return Un("not", e) if negated else e
            pat = self.next()
            return LikeE(left, pat.value, negated)
        if t and t.kind == "id" and t.value.lower() == "ilike":
            self.i += 1
            pat = self.next()
            return LikeE(left, pat.value, negated, ci=True)
        if t and t.kind == "id" and t.value.lower() in ("rlike", "regexp"):
            self.i += 1
            pat = self.next()
            e = Func("regexp_like", [left, Lit(pat.value, "str")])
            return Un("not", e) if negated else e
        if t and t.kind == "kw" and t.value == "is":
            self.i += 1
            neg = bool(self.accept_kw("not"))
            if self.accept_kw("distinct"):
                self.expect_kw("from")
                other = self.parse_add()
                eqn = Func("equal_null", [left, other])
                # IS DISTINCT FROM = NOT equal_null
                return eqn if neg else Un("not", eqn)
            self.expect_kw("null")
            return IsNullE(left, neg)
        return left

    def parse_add(self):
        left = self.parse_mul()
        while True:
            t = self.peek()
            if t and t.kind == "op" and t.value in ("+", "-"):
                self.i += 1
                left = Bin("add" if t.value == "+" else "sub", left,
                           self.parse_mul())
            elif t and t.kind == "op" and t.value == "||":
                self.i += 1
                left = Bin("concat", left, self.parse_mul())
            else:
                return left

    def parse_mul(self):
        left = self.parse_unary()
        while True:
            t = self.peek()
            if t and t.kind == "op" and t.value in ("*", "/", "%"):
                self.i += 1
                op = {"*": "mul", "/": "div", "%": "mod"}[t.value]
                left = Bin(op, left, self.parse_unary())
            else:
                return left

    def parse_unary(self):
        if self.accept_op("-"):
            return Bin("sub", Lit(0, "num"), self.parse_unary())
        if self.accept_op("+"):
            return self.parse_unary()
        return self.parse_primary()

    def parse_primary(self):
        t = self.next()
        if t.kind == "num":
            v = float(t.value) if "." in t.value else int(t.value)
            return Lit(v, "num")
        if t.kind == "str":
            return Lit(t.value, "str")
        if t.kind == "kw":
            if t.value == "date":
                s = self.next()
                return Lit(s.value, "date")
            if t.value == "interval":
                s = self.next()  # '3' or '3 month'
                unit_t = self.peek()
                unit = None
                if unit_t and unit_t.kind == "id":
                    unit = self.next().value.lower()
                parts = s.value.split()
                qty = int(parts[0])
                if unit is None and len(parts) > 1:
                    unit = parts[1].lower()
                return Lit((qty, (unit or "day").rstrip("s")), "interval")
            if t.value == "null":
                return Lit(None, "null")
            if t.value in ("true", "false"):
                return Lit(t.value == "true", "bool")
            if t.value == "case":
                # simple form: CASE <expr> WHEN v THEN r ... rewrites each
                # WHEN into an equality against the operand
                operand = None
                nt = self.peek()
                if not (nt and nt.kind == "kw" and nt.value == "when"):
                    operand = self.parse_expr()
                whens = []
                els = None
                while self.accept_kw("when"):
                    c = self.parse_expr()
                    if operand is not None:
                        c = Bin("eq", operand, c)
                    self.expect_kw("then")
                    v = self.parse_expr()
                    whens.append((c, v))
                if self.accept_kw("else"):
                    els = self.parse_expr()
                self.expect_kw("end")
                return CaseE(whens, els)
            if t.value == "cast":
                self.expect_op("(")
                e = self.parse_expr()
                self.expect_kw("as")
                ty = self.next().value.lower()
                # swallow precision args: decimal(12,2)
                if self.accept_op("("):
                    while not self.accept_op(")"):
                        self.next()
                self.expect_op(")")
                return CastE(e, ty)
            if t.value == "extract":
                self.expect_op("(")
                fld = self.next().value.lower()
                self.expect_kw("from")
                e = self.parse_expr()
                self.expect_op(")")
                return ExtractE(fld, e)
            if t.value == "substring":
                # SUBSTRING(x FROM s [FOR n]) and SUBSTRING(x, s[, n])
                self.expect_op("(")
                e = self.parse_expr()
                length = None
                if self.accept_op(","):
                    start = self.parse_expr()
                    if self.accept_op(","):
                        length = self.parse_expr()
                else:
                    self.expect_kw("from")
                    start = self.parse_expr()
                    if self.accept_kw("for"):
                        length = self.parse_expr()
                self.expect_op(")")
                return Func("substring", [e, start, length])
            if t.value == "exists":
                self.expect_op("(")
                q = self.parse_select()
                self.expect_op(")")
                return ExistsE(q)
            if t.value in ("left", "right") and self.peek() \
                    and self.peek().kind == "op" and self.peek().value == "(":
                # LEFT(s, n) / RIGHT(s, n) string functions (the words are
                # otherwise join keywords)
                self.i += 1
                args = [self.parse_expr()]
                while self.accept_op(","):
                    args.append(self.parse_expr())
                self.expect_op(")")
                return Func(t.value, args)
            raise SyntaxError(f"unexpected keyword {t.value!r}")
        if t.kind == "op" and t.value == "(":
            nt = self.peek()
            if nt and nt.kind == "kw" and nt.value == "select":
                q = self.parse_select()
                self.expect_op(")")
                return SubqueryE(q)
            e = self.parse_expr()
            self.expect_op(")")
            return e
        if t.kind == "id":
            nt = self.peek()
            if nt and nt.kind == "op" and nt.value == "(" \
                    and t.value.lower() in ("try_cast", "try_to_number",
                                            "try_to_double"):
                self.i += 1
                e = self.parse_expr()
                ty = "double"
                if self.accept_kw("as"):
                    ty = self.next().value.lower()
                    if self.accept_op("("):
                        while not self.accept_op(")"):
                            self.next()
                self.expect_op(")")
                return CastE(e, ty, safe=True)
            if nt and nt.kind == "op" and nt.value == "(":
                self.i += 1
                distinct = bool(self.accept_kw("distinct"))
                star = False
                args = []
                if self.accept_op("*"):
                    self.expect_op(")")
                    star = True
                elif t.value.lower() == "position":
                    # POSITION(sub IN str) — parse below IN-predicate level
                    # so the keyword separates the args; the comma form
                    # POSITION(sub, str) also lands here
                    args.append(self.parse_add())
                    if self.accept_kw("in") or self.accept_op(","):
                        args.append(self.parse_expr())
                    self.expect_op(")")
                elif not self.accept_op(")"):
                    args.append(self.parse_expr())
                    while self.accept_op(","):
                        args.append(self.parse_expr())
                    self.expect_op(")")
                ig = self.peek()
                ignore_nulls = False
                if ig and ig.kind == "id" and ig.value.lower() in (
                        "ignore", "respect"):
                    self.i += 1
                    self.next()  # NULLS
                    ignore_nulls = ig.value.lower() == "ignore"
                wt = self.peek()
                if wt and wt.kind == "id" and wt.value.lower() == "within":
                    # fn(args) WITHIN GROUP (ORDER BY e [ASC|DESC])
                    self.i += 1
                    self.expect_kw("group")
                    self.expect_op("(")
                    self.expect_kw("order")
                    self.expect_kw("by")
                    oe = self.parse_expr()
                    asc = True
                    if self.accept_kw("desc"):
                        asc = False
                    else:
                        self.accept_kw("asc")
                    self.expect_op(")")
                    f = Func(t.value.lower(), args, distinct=distinct)
                    f.within_order = (oe, asc)
                    return f
                if self.accept_kw("over"):
                    self.expect_op("(")
                    part, order = [], []
                    if self.accept_kw("partition"):
                        self.expect_kw("by")
                        part.append(self.parse_expr())
                        while self.accept_op(","):
                            part.append(self.parse_expr())
                    if self.accept_kw("order"):
                        self.expect_kw("by")
                        while True:
                            e = self.parse_expr()
                            asc = True
                            if self.accept_kw("desc"):
                                asc = False
                            else:
                                self.accept_kw("asc")
                            order.append((e, asc))
                            if not self.accept_op(","):
                                break
                    frame = None
                    nt = self.peek()
                    if nt and nt.kind == "id" and nt.value.lower() == "rows":
                        self.i += 1
                        self.expect_kw("between")
                        ft = self.next()
                        if ft.kind == "id" and ft.value.lower() == "unbounded":
                            self.next()  # PRECEDING
                            frame = -1
                        else:
                            frame = int(ft.value)
                            self.next()  # PRECEDING
                        self.expect_kw("and")
                        cur = self.next()  # CURRENT | UNBOUNDED
                        if cur.value.lower() == "unbounded":
                            self.next()  # FOLLOWING: whole-partition frame
                            frame = None if frame == -1 else frame
                            # start=UNBOUNDED end=UNBOUNDED -> no frame
                            # (partition-wide); bounded start keeps trailing
                        else:
                            assert cur.value.lower() == "current", cur
                            self.next()  # ROW
                    self.expect_op(")")
                    w = WindowE(t.value.lower(), args, part, order,
                                star=star, frame_preceding=frame,
                                distinct=distinct)
                    w.ignore_nulls = ignore_nulls
                    return w
                if star:
                    return Func(t.value.lower(), [], star=True)
                return Func(t.value.lower(), args, distinct=distinct)
            if nt and nt.kind == "op" and nt.value == ".":
                self.i += 1
                col = self.next()
                return Col(t.value, col.value)
            return Col(None, t.value)
        raise SyntaxError(f"unexpected token {t}")
