"""SQL tokenizer + recursive-descent parser.

Covers the analytic dialect the reference's SQL frontend supports for TPC-H
(/root/reference/src/daft-sql/src/planner.rs): SELECT with expressions /
aggregates / CASE / EXTRACT / SUBSTRING, FROM with comma joins and
JOIN..ON, WHERE/GROUP BY/HAVING/ORDER BY/LIMIT, CTEs, derived tables,
scalar/IN/EXISTS subqueries, DATE and INTERVAL literals.
"""
from __future__ import annotations

import re
from dataclasses import dataclass, field
from typing import Any, List, Optional, Tuple

KEYWORDS = {
    "select", "from", "where", "group", "by", "having", "order", "limit",
    "offset", "as", "and", "or", "not", "in", "exists", "between", "like",
    "ilike", "is", "null", "case", "when", "then", "else", "end", "join",
    "inner", "left", "right", "full", "outer", "cross", "on", "union",
    "all", "distinct", "asc", "desc", "interval", "date", "extract",
    "substring", "for", "with", "count", "sum", "avg", "min", "max",
    "first", "last", "nulls", "semi", "anti", "true", "false", "cast",
    "over", "partition", "rows", "between", "unbounded", "preceding",
    "intersect", "except",
    "current", "row", "following",
}

TOKEN_RE = re.compile(r"""
    (?P<ws>\s+|--[^\n]*)
  | (?P<num>\d+\.\d*|\.\d+|\d+)
  | (?P<str>'(?:[^']|'')*')
  | (?P<name>[A-Za-z_][A-Za-z_0-9]*)
  | (?P<op><>|!=|>=|<=|\|\||[-+*/%(),.<>=])
""", re.VERBOSE)


@dataclass
class Token:
    kind: str  # num | str | name | kw | op | eof
    value: str
    pos: int


def tokenize(text: str) -> List[Token]:
    out: List[Token] = []
    pos = 0
    n = len(text)
    while pos < n:
        m = TOKEN_RE.match(text, pos)
        if not m:
            raise SQLParseError(f"unexpected character {text[pos]!r} at {pos}")
        pos = m.end()
        if m.lastgroup == "ws":
            continue
        kind = m.lastgroup
        val = m.group()
        if kind == "name" and val.lower() in KEYWORDS:
            out.append(Token("kw", val.lower(), m.start()))
        elif kind == "str":
            out.append(Token("str", val[1:-1].replace("''", "'"), m.start()))
        else:
            out.append(Token(kind, val, m.start()))
    out.append(Token("eof", "", n))
    return out


class SQLParseError(ValueError):
    pass


# ---------------------------------------------------------------------------
# AST
# ---------------------------------------------------------------------------

@dataclass
class Col:
    table: Optional[str]
    name: str


@dataclass
class Lit:
    value: Any


@dataclass
class DateLit:
    value: str


@dataclass
class IntervalLit:
    n: int
    unit: str  # day | month | year


@dataclass
class BinOp:
    op: str
    left: Any
    right: Any


@dataclass
class UnaryOp:
    op: str  # not | neg
    child: Any


@dataclass
class FuncCall:
    name: str
    args: List[Any]
    distinct: bool = False
    star: bool = False


@dataclass
class SetOpStmt:
    op: str                 # union | intersect | except
    all: bool
    left: Any
    right: Any
    order_by: Any = None
    limit: Any = None
    offset: int = 0


@dataclass
class WindowExpr:
    func: Any                       # FuncCall underneath
    partition_by: List[Any]
    order_by: List[Tuple[Any, bool]]  # (expr, desc)
    frame: Any = None               # (lo, hi) row offsets or None


@dataclass
class CaseExpr:
    whens: List[Tuple[Any, Any]]
    default: Optional[Any]


@dataclass
class CastExpr:
    child: Any
    type_name: str


@dataclass
class BetweenExpr:
    child: Any
    lo: Any
    hi: Any
    negated: bool = False


@dataclass
class InList:
    child: Any
    values: List[Any]
    negated: bool = False


@dataclass
class LikeExpr:
    child: Any
    pattern: str
    negated: bool = False
    case_insensitive: bool = False


@dataclass
class IsNullExpr:
    child: Any
    negated: bool = False


@dataclass
class SubqueryExpr:
    query: "SelectStmt"


@dataclass
class InSubquery:
    child: Any
    query: "SelectStmt"
    negated: bool = False


@dataclass
class ExistsExpr:
    query: "SelectStmt"
    negated: bool = False


@dataclass
class ExtractExpr:
    part: str
    child: Any


@dataclass
class SubstringExpr:
    child: Any
    start: Any
    length: Optional[Any]


@dataclass
class TableRef:
    name: Optional[str] = None           # base table
    subquery: Optional["SelectStmt"] = None
    alias: Optional[str] = None
    values: Optional[List[List[Any]]] = None   # VALUES rows (expr ASTs)
    col_names: Optional[List[str]] = None      # AS v(c1, c2) column list
    fn: Optional[str] = None                   # table function name
    fn_args: Optional[List[Any]] = None


@dataclass
class JoinClause:
    table: TableRef
    how: str                 # inner/left/right/outer/cross/semi/anti
    on: Optional[Any]


@dataclass
class SelectItem:
    expr: Any
    alias: Optional[str]
    star: bool = False


@dataclass
class OrderItem:
    expr: Any
    desc: bool = False
    nulls_first: Optional[bool] = None


@dataclass
class SelectStmt:
    ctes: List[Tuple[str, "SelectStmt"]] = field(default_factory=list)
    distinct: bool = False
    items: List[SelectItem] = field(default_factory=list)
    from_tables: List[TableRef] = field(default_factory=list)
    joins: List[JoinClause] = field(default_factory=list)
    where: Optional[Any] = None
    group_by: List[Any] = field(default_factory=list)
    having: Optional[Any] = None
    qualify: Optional[Any] = None
    order_by: List[OrderItem] = field(default_factory=list)
    limit: Optional[int] = None
    offset: Optional[int] = None


# ---------------------------------------------------------------------------
# parser
# ---------------------------------------------------------------------------

class Parser:
    def __init__(self, text: str):
        self.toks = tokenize(text)
        self.i = 0

    # -- token helpers ---------------------------------------------------
    def peek(self, k: int = 0) -> Token:
        return self.toks[min(self.i + k, len(self.toks) - 1)]

    def next(self) -> Token:
        t = self.toks[self.i]
        self.i += 1
        return t

    def at_kw(self, *kws: str) -> bool:
        t = self.peek()
        return t.kind == "kw" and t.value in kws

    def eat_kw(self, *kws: str) -> bool:
        if self.at_kw(*kws):
            self.next()
            return True
        return False

    def expect_kw(self, kw: str):
        if not self.eat_kw(kw):
            raise SQLParseError(f"expected {kw.upper()} at {self.peek()}")

    def at_op(self, *ops: str) -> bool:
        t = self.peek()
        return t.kind == "op" and t.value in ops

    def eat_op(self, *ops: str) -> bool:
        if self.at_op(*ops):
            self.next()
            return True
        return False

    def expect_op(self, op: str):
        if not self.eat_op(op):
            raise SQLParseError(f"expected {op!r}, got {self.peek()}")

    # -- entry -----------------------------------------------------------
    def parse_statement(self) -> SelectStmt:
        stmt = self._parse_set_expr()
        if self.peek().kind != "eof":
            raise SQLParseError(f"trailing tokens at {self.peek()}")
        return stmt

    def _combine_setop(self, op: str, is_all: bool, left, right):
        # ORDER BY / LIMIT written after the set op parse into the
        # right-hand select; they bind to the combined result
        order_by, limit, offset = right.order_by, right.limit, \
            getattr(right, "offset", 0)
        right.order_by, right.limit = [], None
        if hasattr(right, "offset"):
            right.offset = 0
        return SetOpStmt(op, is_all, left, right, order_by, limit, offset)

    def _parse_set_expr(self) -> SelectStmt:
        # standard SQL precedence: INTERSECT binds tighter than UNION/EXCEPT
        stmt = self._parse_intersect_expr()
        while self.at_kw("union", "except"):
            op = self.next().value.lower()
            is_all = self.eat_kw("all")
            right = self._parse_intersect_expr()
            stmt = self._combine_setop(op, is_all, stmt, right)
        return stmt

    def _parse_intersect_expr(self) -> SelectStmt:
        stmt = self._parse_set_operand()
        while self.at_kw("intersect"):
            self.next()
            is_all = self.eat_kw("all")
            right = self._parse_set_operand()
            stmt = self._combine_setop("intersect", is_all, stmt, right)
        return stmt

    def _parse_set_operand(self) -> SelectStmt:
        # parenthesized query: "(" SELECT/WITH ... ")"
        if self.at_op("(") and self.peek(1).kind == "kw" and \
                self.peek(1).value in ("select", "with"):
            self.next()
            inner = self._parse_set_expr()
            self.expect_op(")")
            return inner
        return self.parse_select()

    def parse_select(self) -> SelectStmt:
        s = SelectStmt()
        if self.eat_kw("with"):
            while True:
                name = self.next().value
                self.expect_kw("as")
                self.expect_op("(")
                sub = self.parse_select()
                self.expect_op(")")
                s.ctes.append((name, sub))
                if not self.eat_op(","):
                    break
        self.expect_kw("select")
        if self.eat_kw("distinct"):
            s.distinct = True
        while True:
            s.items.append(self.parse_select_item())
            if not self.eat_op(","):
                break
        # SELECT without FROM: constant projection over one dummy row
        if self.eat_kw("from"):
            s.from_tables.append(self.parse_table_ref())
            while True:
                if self.eat_op(","):
                    s.from_tables.append(self.parse_table_ref())
                    continue
                how = self._try_join_kind()
                if how is None:
                    break
                tref = self.parse_table_ref()
                on = None
                if self.eat_kw("on"):
                    on = self.parse_expr()
                s.joins.append(JoinClause(tref, how, on))
        if self.eat_kw("where"):
            s.where = self.parse_expr()
        if self.eat_kw("group"):
            self.expect_kw("by")
            if self.at_kw("all") or (self.peek().kind == "name" and
                                     self.peek().value.lower() == "all"):
                self.next()
                s.group_by.append("__GROUP_BY_ALL__")
            else:
                while True:
                    s.group_by.append(self.parse_expr())
                    if not self.eat_op(","):
                        break
        if self.eat_kw("having"):
            s.having = self.parse_expr()
        if self.peek().kind == "name" and \
                self.peek().value.lower() == "qualify":
            self.next()
            s.qualify = self.parse_expr()
        if self.eat_kw("order"):
            self.expect_kw("by")
            while True:
                e = self.parse_expr()
                desc = False
                if self.eat_kw("desc"):
                    desc = True
                elif self.eat_kw("asc"):
                    pass
                nf = None
                if self.eat_kw("nulls"):
                    nf = self.eat_kw("first")
                    if not nf:
                        self.expect_kw("last")
                s.order_by.append(OrderItem(e, desc, nf))
                if not self.eat_op(","):
                    break
        if self.eat_kw("limit"):
            s.limit = int(self.next().value)
        if self.eat_kw("offset"):
            s.offset = int(self.next().value)
        return s

    def _try_join_kind(self) -> Optional[str]:
        if self.eat_kw("join"):
            return "inner"
        for kw, how in (("inner", "inner"), ("left", "left"),
                        ("right", "right"), ("full", "outer"),
                        ("cross", "cross"), ("semi", "semi"),
                        ("anti", "anti")):
            if self.at_kw(kw):
                self.next()
                self.eat_kw("outer")
                self.expect_kw("join")
                return how
        return None

    def parse_select_item(self) -> SelectItem:
        if self.at_op("*"):
            self.next()
            return SelectItem(None, None, star=True)
        e = self.parse_expr()
        alias = None
        if self.eat_kw("as"):
            alias = self.next().value
        elif self.peek().kind == "name":
            alias = self.next().value
        return SelectItem(e, alias)

    def parse_table_ref(self) -> TableRef:
        if self.eat_op("("):
            if self._at_name("values"):
                tr = self._parse_values()
            else:
                tr = TableRef(subquery=self.parse_select())
            self.expect_op(")")
            self._parse_table_alias(tr)
            return tr
        if self._at_name("values"):
            tr = self._parse_values()
            self._parse_table_alias(tr)
            return tr
        name = self.next().value
        if self.at_op("("):
            # table function: read_parquet('path'), read_csv(...), ...
            self.next()
            args = []
            if not self.at_op(")"):
                while True:
                    args.append(self.parse_expr())
                    if not self.eat_op(","):
                        break
            self.expect_op(")")
            tr = TableRef(fn=name, fn_args=args)
            self._parse_table_alias(tr)
            return tr
        tr = TableRef(name=name)
        self._parse_table_alias(tr)
        return tr

    def _at_name(self, word: str) -> bool:
        t = self.peek()
        return (t.kind == "name" and t.value.lower() == word) or \
            (t.kind == "kw" and t.value == word)

    def _parse_values(self) -> TableRef:
        self.next()                      # VALUES
        rows = []
        while True:
            self.expect_op("(")
            row = []
            while True:
                row.append(self.parse_expr())
                if not self.eat_op(","):
                    break
            self.expect_op(")")
            rows.append(row)
            if not self.eat_op(","):
                break
        return TableRef(values=rows)

    def _parse_table_alias(self, tr: TableRef):
        if self.eat_kw("as"):
            tr.alias = self.next().value
        elif self.peek().kind == "name" and \
                self.peek().value.lower() not in ("values", "qualify"):
            tr.alias = self.next().value
        else:
            return
        if self.eat_op("("):
            names = []
            while True:
                names.append(self.next().value)
                if not self.eat_op(","):
                    break
            self.expect_op(")")
            tr.col_names = names

    # -- expressions (precedence climbing) --------------------------------
    def parse_expr(self) -> Any:
        return self.parse_or()

    def parse_or(self) -> Any:
        left = self.parse_and()
        while self.eat_kw("or"):
            left = BinOp("or", left, self.parse_and())
        return left

    def parse_and(self) -> Any:
        left = self.parse_not()
        while self.eat_kw("and"):
            left = BinOp("and", left, self.parse_not())
        return left

    def parse_not(self) -> Any:
        if self.eat_kw("not"):
            return UnaryOp("not", self.parse_not())
        return self.parse_predicate()

    def parse_predicate(self) -> Any:
        left = self.parse_addsub()
        negated = False
        if self.at_kw("not"):
            nxt = self.peek(1)
            if nxt.kind == "kw" and nxt.value in ("in", "between", "like",
                                                  "ilike"):
                self.next()
                negated = True
        if self.eat_kw("between"):
            lo = self.parse_addsub()
            self.expect_kw("and")
            hi = self.parse_addsub()
            return BetweenExpr(left, lo, hi, negated)
        if self.eat_kw("in"):
            self.expect_op("(")
            if self.at_kw("select", "with"):
                sub = self.parse_select()
                self.expect_op(")")
                return InSubquery(left, sub, negated)
            vals = [self.parse_expr()]
            while self.eat_op(","):
                vals.append(self.parse_expr())
            self.expect_op(")")
            return InList(left, vals, negated)
        if self.at_kw("like", "ilike"):
            ci = self.next().value == "ilike"
            pat = self.next()
            if pat.kind != "str":
                raise SQLParseError("LIKE pattern must be a string literal")
            return LikeExpr(left, pat.value, negated, ci)
        if self.eat_kw("is"):
            neg = self.eat_kw("not")
            self.expect_kw("null")
            return IsNullExpr(left, neg)
        for op_tok, op in (("=", "eq"), ("<>", "ne"), ("!=", "ne"),
                           ("<=", "le"), (">=", "ge"), ("<", "lt"),
                           (">", "gt")):
            if self.at_op(op_tok):
                self.next()
                right = self.parse_addsub()
                return BinOp(op, left, right)
        return left

    def parse_addsub(self) -> Any:
        left = self.parse_muldiv()
        while True:
            if self.eat_op("+"):
                left = BinOp("add", left, self.parse_muldiv())
            elif self.eat_op("-"):
                left = BinOp("sub", left, self.parse_muldiv())
            elif self.eat_op("||"):
                left = BinOp("concat", left, self.parse_muldiv())
            else:
                return left

    def parse_muldiv(self) -> Any:
        left = self.parse_unary()
        while True:
            if self.eat_op("*"):
                left = BinOp("mul", left, self.parse_unary())
            elif self.eat_op("/"):
                left = BinOp("div", left, self.parse_unary())
            elif self.eat_op("%"):
                left = BinOp("mod", left, self.parse_unary())
            else:
                return left

    def parse_unary(self) -> Any:
        if self.eat_op("-"):
            return UnaryOp("neg", self.parse_unary())
        if self.eat_op("+"):
            return self.parse_unary()
        return self.parse_primary()

    def parse_primary(self) -> Any:
        t = self.peek()
        if t.kind == "op" and t.value == "(":
            self.next()
            if self.at_kw("select", "with"):
                sub = self.parse_select()
                self.expect_op(")")
                return SubqueryExpr(sub)
            e = self.parse_expr()
            self.expect_op(")")
            return e
        if t.kind == "num":
            self.next()
            v = float(t.value) if ("." in t.value) else int(t.value)
            return Lit(v)
        if t.kind == "str":
            self.next()
            return Lit(t.value)
        if t.kind == "kw":
            if t.value == "true":
                self.next()
                return Lit(True)
            if t.value == "false":
                self.next()
                return Lit(False)
            if t.value == "null":
                self.next()
                return Lit(None)
            if t.value == "date":
                self.next()
                d = self.next()
                return DateLit(d.value)
            if t.value == "interval":
                self.next()
                body = self.next().value  # e.g. '3' or '3 month'
                unit = "day"
                parts = body.split()
                n = int(parts[0].strip("'"))
                if len(parts) > 1:
                    unit = parts[1].rstrip("s").lower()
                elif self.peek().kind == "name" or self.peek().kind == "kw":
                    unit = self.next().value.rstrip("s").lower()
                return IntervalLit(n, unit)
            if t.value == "case":
                self.next()
                whens = []
                default = None
                while self.eat_kw("when"):
                    c = self.parse_expr()
                    self.expect_kw("then")
                    v = self.parse_expr()
                    whens.append((c, v))
                if self.eat_kw("else"):
                    default = self.parse_expr()
                self.expect_kw("end")
                return CaseExpr(whens, default)
            if t.value == "extract":
                self.next()
                self.expect_op("(")
                part = self.next().value.lower()
                self.expect_kw("from")
                child = self.parse_expr()
                self.expect_op(")")
                return ExtractExpr(part, child)
            if t.value == "substring":
                self.next()
                self.expect_op("(")
                child = self.parse_expr()
                if self.eat_kw("from"):
                    start = self.parse_expr()
                    length = None
                    if self.eat_kw("for"):
                        length = self.parse_expr()
                else:
                    self.expect_op(",")
                    start = self.parse_expr()
                    length = None
                    if self.eat_op(","):
                        length = self.parse_expr()
                self.expect_op(")")
                return SubstringExpr(child, start, length)
            if t.value == "cast":
                self.next()
                self.expect_op("(")
                child = self.parse_expr()
                self.expect_kw("as")
                tn = self.next().value.lower()
                while self.peek().kind == "name":
                    tn += " " + self.next().value.lower()
                if self.eat_op("("):
                    args = []
                    while not self.eat_op(")"):
                        tok = self.next()
                        if tok.value != ",":
                            args.append(str(tok.value))
                    tn += "(" + ",".join(args) + ")"
                self.expect_op(")")
                return CastExpr(child, tn)
            if t.value == "exists":
                self.next()
                self.expect_op("(")
                sub = self.parse_select()
                self.expect_op(")")
                return ExistsExpr(sub)
            if t.value in ("count", "sum", "avg", "min", "max"):
                return self._parse_func_like()
            if t.value == "not":
                self.next()
                return UnaryOp("not", self.parse_primary())
        if t.kind == "name":
            nxt = self.peek(1)
            if nxt.kind == "op" and nxt.value == "(":
                return self._parse_func_like()
            self.next()
            if self.at_op(".") and self.peek(1).kind in ("name", "kw"):
                self.next()
                col = self.next().value
                return Col(t.value, col)
            return Col(None, t.value)
        raise SQLParseError(f"unexpected token {t}")

    def _parse_func_like(self) -> Any:
        name = self.next().value.lower()
        self.expect_op("(")
        if self.at_op("*"):
            self.next()
            self.expect_op(")")
            return FuncCall(name, [], star=True)
        distinct = self.eat_kw("distinct")
        args = []
        if not self.at_op(")"):
            args.append(self.parse_expr())
            while self.eat_op(","):
                args.append(self.parse_expr())
        self.expect_op(")")
        fn = FuncCall(name, args, distinct=distinct)
        if self.at_kw("over"):
            return self._parse_over(fn)
        return fn

    def _parse_over(self, fn):
        self.expect_kw("over")
        self.expect_op("(")
        parts, order, frame = [], [], None
        if self.eat_kw("partition"):
            self.expect_kw("by")
            parts.append(self.parse_expr())
            while self.eat_op(","):
                parts.append(self.parse_expr())
        if self.eat_kw("order"):
            self.expect_kw("by")
            while True:
                e = self.parse_expr()
                desc = False
                if self.eat_kw("desc"):
                    desc = True
                elif self.eat_kw("asc"):
                    pass
                order.append((e, desc))
                if not self.eat_op(","):
                    break
        if self.eat_kw("rows"):
            self.expect_kw("between")

            def bound():
                if self.eat_kw("unbounded"):
                    side = self.next().value.lower()
                    return None if side == "preceding" else None
                if self.eat_kw("current"):
                    self.expect_kw("row")
                    return 0
                n = int(self.next().value)
                side = self.next().value.lower()
                return -n if side == "preceding" else n
            lo = bound()
            self.expect_kw("and")
            hi = bound()
            frame = (lo, hi)
        self.expect_op(")")
        return WindowExpr(fn, parts, order, frame)


def parse_sql(text: str) -> SelectStmt:
    return Parser(text).parse_statement()


def parse_expression(text: str):
    """Scalar SQL expression -> daft_amd Expression (unbound columns)."""
    p = Parser(text)
    ast = p.parse_expr()
    if p.peek().kind != "eof":
        raise SQLParseError(f"trailing tokens at {p.peek()}")
    from .planner import expr_to_daft
    return expr_to_daft(ast, None)
