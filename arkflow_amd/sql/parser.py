"""SQL frontend: lexer + recursive-descent parser for the engine's SQL subset.

Replaces DataFusion's SQL layer (reference processor/sql.rs:185-201 parses with
sqlparser-rs and plans via DataFusion). Supported subset — the surface that
ArkFlow pipelines actually use over table ``flow``:

  SELECT [DISTINCT] expr [AS alias], ...
  FROM table [ [INNER|LEFT] JOIN table ON a.x = b.y ]...
  [WHERE pred] [GROUP BY exprs] [HAVING pred]
  [ORDER BY expr [ASC|DESC], ...] [LIMIT n]

Expressions: arithmetic, comparison, AND/OR/NOT, BETWEEN, IN (list),
IS [NOT] NULL, LIKE (prefix/suffix/contains), CASE WHEN, CAST(expr AS type),
unary +/-, scalar functions, aggregate functions (COUNT/SUM/AVG/MIN/MAX),
string and numeric literals. DDL/DML is rejected, matching the reference's
SQLOptions (processor/sql.rs:189-192).
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional, Tuple

from ..errors import ConfigError


class SqlError(ConfigError):
    pass


# ------------------------------------------------------------------------ AST
@dataclass
class ColumnRef:
    name: str
    table: Optional[str] = None


@dataclass
class Literal:
    value: object  # int | float | str | bool | None


@dataclass
class BinaryOp:
    op: str  # + - * / % = != < <= > >= and or
    left: object
    right: object


@dataclass
class UnaryOp:
    op: str  # - not
    operand: object


@dataclass
class WindowSpec:
    partition_by: list
    order_by: list  # [(expr, asc)]


@dataclass
class FuncCall:
    name: str
    args: list
    distinct: bool = False
    over: "WindowSpec" = None
    filter: object = None  # aggregate FILTER (WHERE ...) condition


@dataclass
class Star:
    table: Optional[str] = None


@dataclass
class Cast:
    expr: object
    to_type: str


@dataclass
class Case:
    whens: List[Tuple[object, object]]
    else_: Optional[object] = None


@dataclass
class Between:
    expr: object
    low: object
    high: object
    negated: bool = False


@dataclass
class InList:
    expr: object
    items: list
    negated: bool = False
    subquery: object = None  # IN (SELECT ...): uncorrelated subselect


@dataclass
class ScalarSubquery:
    select: object  # uncorrelated (SELECT ...) producing one value


@dataclass
class ExistsSubquery:
    select: object
    negated: bool = False


@dataclass
class IsNull:
    expr: object
    negated: bool = False


@dataclass
class Like:
    expr: object
    pattern: str
    negated: bool = False


@dataclass
class Join:
    kind: str  # inner | left
    table: str
    alias: Optional[str]
    on: object


@dataclass
class Select:
    projections: List[Tuple[object, Optional[str]]] = field(default_factory=list)
    distinct: bool = False
    from_table: Optional[str] = None
    from_alias: Optional[str] = None
    joins: List[Join] = field(default_factory=list)
    where: Optional[object] = None
    group_by: list = field(default_factory=list)
    having: Optional[object] = None
    order_by: List[Tuple[object, bool]] = field(default_factory=list)  # (expr, asc)
    limit: Optional[int] = None
    offset: int = 0
    union_all: Optional["Select"] = None  # legacy single UNION ALL chain
    set_ops: list = field(default_factory=list)  # [("union"|"union_all"|"intersect"|"except", Select)]


AGGREGATE_FUNCS = {"count", "sum", "avg", "min", "max"}

# ---------------------------------------------------------------------- lexer
_KEYWORDS = {
    "select", "distinct", "from", "where", "group", "by", "having", "order",
    "limit", "as", "and", "or", "not", "in", "is", "null", "between", "like",
    "case", "when", "then", "else", "end", "cast", "join", "inner", "left",
    "right", "full", "outer", "on", "asc", "desc", "true", "false",
    "offset", "union", "all", "over", "partition", "row_number", "rank",
    "dense_rank", "nulls", "first", "last",
    # rejected verbs (detected for a clear error)
    "insert", "update", "delete", "create", "drop", "alter", "truncate",
}

_TWO_CHAR = {"<=", ">=", "!=", "<>", "||"}
_ONE_CHAR = set("+-*/%(),.=<>")


@dataclass
class Token:
    kind: str  # kw | ident | number | string | op | end
    value: str
    pos: int


def tokenize(sql: str) -> List[Token]:
    toks: List[Token] = []
    i, n = 0, len(sql)
    while i < n:
        c = sql[i]
        if c.isspace():
            i += 1
            continue
        if c == "-" and i + 1 < n and sql[i + 1] == "-":  # line comment
            while i < n and sql[i] != "\n":
                i += 1
            continue
        if c.isalpha() or c == "_":
            j = i
            while j < n and (sql[j].isalnum() or sql[j] == "_"):
                j += 1
            word = sql[i:j]
            kind = "kw" if word.lower() in _KEYWORDS else "ident"
            toks.append(Token(kind, word.lower() if kind == "kw" else word, i))
            i = j
            continue
        if c.isdigit() or (c == "." and i + 1 < n and sql[i + 1].isdigit()):
            j = i
            seen_dot = False
            while j < n and (sql[j].isdigit() or (sql[j] == "." and not seen_dot)):
                if sql[j] == ".":
                    seen_dot = True
                j += 1
            if j < n and sql[j] in "eE":
                k = j + 1
                if k < n and sql[k] in "+-":
                    k += 1
                while k < n and sql[k].isdigit():
                    k += 1
                j = k
                seen_dot = True
            toks.append(Token("number", sql[i:j], i))
            i = j
            continue
        if c == "'":
            j = i + 1
            buf = []
            while j < n:
                if sql[j] == "'":
                    if j + 1 < n and sql[j + 1] == "'":
                        buf.append("'")
                        j += 2
                        continue
                    break
                buf.append(sql[j])
                j += 1
            if j >= n:
                raise SqlError(f"unterminated string at {i}")
            toks.append(Token("string", "".join(buf), i))
            i = j + 1
            continue
        if c == '"':  # quoted identifier
            j = sql.find('"', i + 1)
            if j < 0:
                raise SqlError(f"unterminated quoted identifier at {i}")
            toks.append(Token("ident", sql[i + 1:j], i))
            i = j + 1
            continue
        if sql[i:i + 2] in _TWO_CHAR:
            toks.append(Token("op", sql[i:i + 2], i))
            i += 2
            continue
        if c in _ONE_CHAR:
            toks.append(Token("op", c, i))
            i += 1
            continue
        raise SqlError(f"unexpected character {c!r} at {i}")
    toks.append(Token("end", "", n))
    return toks


# --------------------------------------------------------------------- parser
class Parser:
    def __init__(self, sql: str):
        self.toks = tokenize(sql)
        self.i = 0

    # navigation
    def peek(self) -> Token:
        return self.toks[self.i]

    def next(self) -> Token:
        t = self.toks[self.i]
        self.i += 1
        return t

    def accept(self, kind: str, value: Optional[str] = None) -> Optional[Token]:
        t = self.peek()
        if t.kind == kind and (value is None or t.value == value):
            return self.next()
        return None

    def expect(self, kind: str, value: Optional[str] = None) -> Token:
        t = self.accept(kind, value)
        if t is None:
            got = self.peek()
            raise SqlError(
                f"expected {value or kind}, got {got.value!r} at {got.pos}")
        return t

    # entry
    def parse(self) -> Select:
        sel = self._parse_select()
        self.expect("end")
        return sel

    def _parse_select(self) -> Select:
        sel = self._parse_select_core()
        return self._parse_compound_tail(sel)

    def _parse_select_core(self) -> Select:
        t = self.peek()
        if t.kind == "kw" and t.value in (
            "insert", "update", "delete", "create", "drop", "alter", "truncate"
        ):
            raise SqlError(f"DDL/DML not allowed: {t.value.upper()}")
        self.expect("kw", "select")
        sel = Select()
        if self.accept("kw", "distinct"):
            sel.distinct = True
        sel.projections = self._projections()
        if self.accept("kw", "from"):
            # derived table: FROM ( SELECT ... ) [AS] alias
            if self.accept("op", "("):
                sel.from_table = self._parse_select()
                self.expect("op", ")")
            else:
                sel.from_table = self._table_name()
            alias = self._maybe_alias()
            sel.from_alias = alias
            while True:
                kind = None
                if self.accept("kw", "inner"):
                    kind = "inner"
                elif self.accept("kw", "left"):
                    self.accept("kw", "outer")
                    kind = "left"
                if self.accept("kw", "join"):
                    kind = kind or "inner"
                    if self.accept("op", "("):
                        tname = self._parse_select()
                        self.expect("op", ")")
                    else:
                        tname = self._table_name()
                    talias = self._maybe_alias()
                    self.expect("kw", "on")
                    on = self._expr()
                    sel.joins.append(Join(kind, tname, talias, on))
                    continue
                if kind is not None:
                    raise SqlError("expected JOIN")
                break
        if self.accept("kw", "where"):
            sel.where = self._expr()
        if self.accept("kw", "group"):
            self.expect("kw", "by")
            sel.group_by.append(self._expr())
            while self.accept("op", ","):
                sel.group_by.append(self._expr())
        if self.accept("kw", "having"):
            sel.having = self._expr()
        return sel

    def _parse_compound_tail(self, sel: Select) -> Select:
        # compound selects: UNION [ALL] / INTERSECT / EXCEPT between the
        # cores; a trailing ORDER BY / LIMIT applies to the WHOLE compound
        # (sqlite/standard)
        while True:
            t = self.peek()
            if t.kind == "kw" and t.value == "union":
                self.next()
                op = "union_all" if self.accept("kw", "all") else "union"
            elif t.kind == "ident" and t.value.lower() in ("intersect",
                                                           "except"):
                self.next()
                op = t.value.lower()
            else:
                break
            sel.set_ops.append((op, self._parse_select_core()))
        if len(sel.set_ops) == 1 and sel.set_ops[0][0] == "union_all":
            sel.union_all = sel.set_ops[0][1]  # legacy field (fast paths)
        if self.accept("kw", "order"):
            self.expect("kw", "by")
            while True:
                e = self._expr()
                asc = True
                if self.accept("kw", "desc"):
                    asc = False
                else:
                    self.accept("kw", "asc")
                nulls_first = None  # default: NULL sorts smallest
                if self.accept("kw", "nulls"):
                    if self.accept("kw", "first"):
                        nulls_first = True
                    else:
                        self.expect("kw", "last")
                        nulls_first = False
                sel.order_by.append((e, asc, nulls_first))
                if not self.accept("op", ","):
                    break
        if self.accept("kw", "limit"):
            sel.limit = int(self.expect("number").value)
        if self.accept("kw", "offset"):
            sel.offset = int(self.expect("number").value)
        return sel

    def _table_name(self) -> str:
        return self.expect("ident").value

    def _maybe_alias(self) -> Optional[str]:
        if self.accept("kw", "as"):
            return self.expect("ident").value
        t = self.peek()
        if t.kind == "ident" and t.value.lower() not in ("intersect",
                                                         "except"):
            return self.next().value
        return None

    def _projections(self):
        projs = []
        while True:
            if self.accept("op", "*"):
                projs.append((Star(), None))
            else:
                save = self.i
                # qualified star: t.*
                t = self.accept("ident")
                if t and self.accept("op", ".") and self.accept("op", "*"):
                    projs.append((Star(t.value), None))
                else:
                    self.i = save
                    e = self._expr()
                    alias = None
                    if self.accept("kw", "as"):
                        alias = self.expect("ident").value
                    elif self.peek().kind == "ident":
                        alias = self.next().value
                    projs.append((e, alias))
            if not self.accept("op", ","):
                break
        return projs

    # expression precedence: or < and < not < cmp < add < mul < unary < postfix
    def _expr(self):
        return self._or()

    def _or(self):
        left = self._and()
        while self.accept("kw", "or"):
            left = BinaryOp("or", left, self._and())
        return left

    def _and(self):
        left = self._not()
        while self.accept("kw", "and"):
            left = BinaryOp("and", left, self._not())
        return left

    def _not(self):
        if self.accept("kw", "not"):
            return UnaryOp("not", self._not())
        return self._cmp()

    def _cmp(self):
        left = self._add()
        t = self.peek()
        if t.kind == "op" and t.value in ("=", "!=", "<>", "<", "<=", ">", ">="):
            self.next()
            op = "!=" if t.value == "<>" else t.value
            return BinaryOp(op, left, self._add())
        negated = False
        if t.kind == "kw" and t.value == "not":
            nxt = self.toks[self.i + 1]
            if nxt.kind == "kw" and nxt.value in ("between", "in", "like"):
                self.next()
                negated = True
                t = self.peek()
        if t.kind == "kw" and t.value == "between":
            self.next()
            low = self._add()
            self.expect("kw", "and")
            high = self._add()
            return Between(left, low, high, negated)
        if t.kind == "kw" and t.value == "in":
            self.next()
            self.expect("op", "(")
            if self.peek().kind == "kw" and self.peek().value == "select":
                sub = self._parse_select()
                self.expect("op", ")")
                return InList(left, [], negated, subquery=sub)
            items = [self._expr()]
            while self.accept("op", ","):
                items.append(self._expr())
            self.expect("op", ")")
            return InList(left, items, negated)
        if t.kind == "kw" and t.value == "like":
            self.next()
            pat = self.expect("string").value
            return Like(left, pat, negated)
        if t.kind == "kw" and t.value == "is":
            self.next()
            neg = bool(self.accept("kw", "not"))
            self.expect("kw", "null")
            return IsNull(left, neg)
        return left

    def _add(self):
        left = self._mul()
        while True:
            t = self.peek()
            if t.kind == "op" and t.value in ("+", "-", "||"):
                self.next()
                left = BinaryOp(t.value, left, self._mul())
            else:
                return left

    def _mul(self):
        left = self._unary()
        while True:
            t = self.peek()
            if t.kind == "op" and t.value in ("*", "/", "%"):
                self.next()
                left = BinaryOp(t.value, left, self._unary())
            else:
                return left

    def _unary(self):
        if self.accept("op", "-"):
            return UnaryOp("-", self._unary())
        if self.accept("op", "+"):
            return self._unary()
        return self._primary()

    def _primary(self):
        t = self.next()
        if t.kind == "number":
            v = float(t.value) if ("." in t.value or "e" in t.value.lower()) \
                else int(t.value)
            return Literal(v)
        if t.kind == "string":
            return Literal(t.value)
        if t.kind == "kw" and t.value in ("true", "false"):
            return Literal(t.value == "true")
        if t.kind == "kw" and t.value == "null":
            return Literal(None)
        if t.kind == "kw" and t.value == "case":
            # simple form `CASE expr WHEN v THEN r ...` desugars to the
            # searched form with `expr = v` conditions (sqlite/standard)
            operand = None
            if self.peek().kind != "kw" or self.peek().value != "when":
                operand = self._expr()
            whens = []
            else_ = None
            while self.accept("kw", "when"):
                cond = self._expr()
                if operand is not None:
                    cond = BinaryOp("=", operand, cond)
                self.expect("kw", "then")
                whens.append((cond, self._expr()))
            if self.accept("kw", "else"):
                else_ = self._expr()
            self.expect("kw", "end")
            return Case(whens, else_)
        if t.kind == "kw" and t.value == "cast":
            self.expect("op", "(")
            e = self._expr()
            self.expect("kw", "as")
            ty = self.expect("ident").value.lower()
            self.expect("op", ")")
            return Cast(e, ty)
        if t.kind == "op" and t.value == "(":
            if self.peek().kind == "kw" and self.peek().value == "select":
                sub = self._parse_select()
                self.expect("op", ")")
                return ScalarSubquery(sub)
            e = self._expr()
            self.expect("op", ")")
            return e
        if t.kind == "ident" and t.value.lower() == "exists":
            self.expect("op", "(")
            sub = self._parse_select()
            self.expect("op", ")")
            return ExistsSubquery(sub)
        if t.kind == "kw" and t.value == "not" \
                and self.peek().kind == "ident" \
                and self.peek().value.lower() == "exists":
            self.next()
            self.expect("op", "(")
            sub = self._parse_select()
            self.expect("op", ")")
            return ExistsSubquery(sub, negated=True)
        if t.kind == "ident" or (t.kind == "kw" and t.value in (
                "left", "right", "row_number", "rank", "dense_rank")):
            name = t.value
            # function call
            if self.accept("op", "("):
                distinct = bool(self.accept("kw", "distinct"))
                args = []
                if self.accept("op", "*"):
                    args.append(Star())
                elif not (self.peek().kind == "op" and self.peek().value == ")"):
                    args.append(self._expr())
                    while self.accept("op", ","):
                        args.append(self._expr())
                self.expect("op", ")")
                filt = None
                t2 = self.peek()
                if t2.kind == "ident" and t2.value.lower() == "filter":
                    self.next()
                    self.expect("op", "(")
                    self.expect("kw", "where")
                    filt = self._expr()
                    self.expect("op", ")")
                over = None
                if self.accept("kw", "over"):
                    self.expect("op", "(")
                    part, order = [], []
                    if self.accept("kw", "partition"):
                        self.expect("kw", "by")
                        part.append(self._expr())
                        while self.accept("op", ","):
                            part.append(self._expr())
                    if self.accept("kw", "order"):
                        self.expect("kw", "by")
                        while True:
                            e = self._expr()
                            asc = True
                            if self.accept("kw", "desc"):
                                asc = False
                            else:
                                self.accept("kw", "asc")
                            order.append((e, asc))
                            if not self.accept("op", ","):
                                break
                    self.expect("op", ")")
                    over = WindowSpec(part, order)
                return FuncCall(name.lower(), args, distinct, over, filt)
            # qualified column
            if self.accept("op", "."):
                col = self.expect("ident").value
                return ColumnRef(col, table=name)
            return ColumnRef(name)
        raise SqlError(f"unexpected token {t.value!r} at {t.pos}")


def parse_sql(sql: str) -> Select:
    return Parser(sql).parse()


def parse_expression(text: str):
    """Parse ONE standalone SQL expression (reference expr/mod.rs Expr{expr}:
    per-row dynamic config values like a kafka output's topic/key)."""
    p = Parser(text)
    e = p._expr()
    if p.peek().kind != "end":
        raise SqlError(f"trailing input in expression: {text!r}")
    return e
