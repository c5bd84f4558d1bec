"""SQL -> DataFrame planner (the analog of the reference's
/root/reference/src/daft-sql/src/planner.rs SQLPlanner).

Features: multi-table FROM with equi-join extraction from WHERE (the TPC-H
comma-join style), explicit JOIN..ON, aggregates with GROUP BY aliases /
ordinals / expressions, HAVING, ORDER BY (aliases/ordinals/exprs), LIMIT,
CTEs, derived tables, CASE/EXTRACT/SUBSTRING/CAST, DATE and INTERVAL
arithmetic (folded at plan time), and subqueries:

  * uncorrelated scalar subquery   -> evaluated, inlined as a literal
  * uncorrelated IN / NOT IN       -> semi / anti join
  * EXISTS / NOT EXISTS with equality correlation -> semi / anti join
  * correlated scalar aggregate with equality correlation -> decorrelated
    into a groupby + join (the Q2/Q17-style rewrite)
"""
from __future__ import annotations

import datetime as _dt
from dataclasses import dataclass
from typing import Any, Callable, Dict, List, Optional, Set, Tuple

from ..expressions.expressions import (Agg, AggKind, Alias, ColumnRef,
                                       Expression, col, lit)
from ..schema import DataType
from . import parser as P


class SQLPlanError(ValueError):
    pass


@dataclass
class Binder:
    """Column resolution scope: alias -> {orig_col -> current_name}."""
    tables: Dict[str, Dict[str, str]]
    outer: Optional["Binder"] = None
    select_aliases: Optional[Dict[str, Any]] = None  # alias -> AST

    def resolve(self, table: Optional[str], name: str) -> Optional[str]:
        if table is not None:
            m = self.tables.get(table)
            if m and name in m:
                return m[name]
            return None
        hits = [m[name] for m in self.tables.values() if name in m]
        if len(hits) > 1:
            raise SQLPlanError(f"ambiguous column {name}")
        return hits[0] if hits else None

    def is_outer_col(self, table: Optional[str], name: str) -> bool:
        if self.resolve(table, name) is not None:
            return False
        b = self.outer
        while b is not None:
            if b.resolve(table, name) is not None:
                return True
            b = b.outer
        return False


def plan_sql(query: str, lookup_table: Callable[[str], Any]):
    stmt = P.parse_sql(query)
    return _plan_stmt(stmt, lookup_table)


def _plan_stmt(stmt, lookup):
    if isinstance(stmt, P.SetOpStmt):
        left = _plan_stmt(stmt.left, lookup)
        right = _plan_stmt(stmt.right, lookup)
        if stmt.op == "union":
            df = left.concat(right) if stmt.all else left.union(right)
        elif stmt.op == "intersect":
            df = left.intersect_all(right) if stmt.all \
                else left.intersect(right)
        else:
            df = left.except_all(right) if stmt.all \
                else left.except_distinct(right)
        if stmt.order_by:
            by = [c.name for c in
                  (o.expr for o in stmt.order_by)
                  if hasattr(c, "name")]
            desc = [o.desc for o in stmt.order_by]
            if len(by) == len(stmt.order_by):
                df = df.sort(by, desc=desc)
        if stmt.limit is not None:
            if stmt.offset:
                df = df.offset(stmt.offset)
            df = df.limit(stmt.limit)
        return df
    return _plan_select(stmt, lookup, outer=None)


# ---------------------------------------------------------------------------
# expression binding
# ---------------------------------------------------------------------------

_AGG_FUNCS = {"sum": AggKind.SUM, "avg": AggKind.MEAN, "min": AggKind.MIN,
              "max": AggKind.MAX, "count": AggKind.COUNT}


def _fold_date(e) -> Optional[_dt.date]:
    if isinstance(e, P.DateLit):
        return _dt.date.fromisoformat(e.value)
    if isinstance(e, P.BinOp) and e.op in ("add", "sub"):
        l = _fold_date(e.left)
        if l is None or not isinstance(e.right, P.IntervalLit):
            return None
        iv = e.right
        sign = 1 if e.op == "add" else -1
        if iv.unit == "day":
            return l + _dt.timedelta(days=sign * iv.n)
        months = iv.n * (12 if iv.unit == "year" else 1) * sign
        y = l.year + (l.month - 1 + months) // 12
        m = (l.month - 1 + months) % 12 + 1
        import calendar
        d = min(l.day, calendar.monthrange(y, m)[1])
        return _dt.date(y, m, d)
    return None


def expr_to_daft(e, binder: Optional[Binder]) -> Expression:
    folded = _fold_date(e)
    if folded is not None:
        return lit(folded)
    if isinstance(e, P.Col):
        if binder is None:
            return col(e.name)
        actual = binder.resolve(e.table, e.name)
        if actual is None:
            raise SQLPlanError(
                f"unknown column {(e.table + '.') if e.table else ''}{e.name}")
        return col(actual)
    if isinstance(e, P.Lit):
        return lit(e.value)
    if isinstance(e, P.DateLit):
        return lit(_dt.date.fromisoformat(e.value))
    if isinstance(e, P.IntervalLit):
        if e.unit == "day":
            return lit(e.n)
        raise SQLPlanError("month/year intervals only combine with date "
                           "literals")
    if isinstance(e, P.BinOp):
        # date column ± INTERVAL 'n month/year' (literal dates folded
        # above; columns need calendar arithmetic)
        if e.op in ("add", "sub") and isinstance(e.right, P.IntervalLit) \
                and e.right.unit in ("month", "year"):
            from ..functions import add_months
            iv = e.right
            months = iv.n * (12 if iv.unit == "year" else 1)
            if e.op == "sub":
                months = -months
            return add_months(expr_to_daft(e.left, binder), months)
        l = expr_to_daft(e.left, binder)
        r = expr_to_daft(e.right, binder)
        ops = {"add": l.__add__, "sub": l.__sub__, "mul": l.__mul__,
               "div": l.__truediv__, "mod": l.__mod__,
               "eq": l.__eq__, "ne": l.__ne__, "lt": l.__lt__,
               "le": l.__le__, "gt": l.__gt__, "ge": l.__ge__,
               "and": l.__and__, "or": l.__or__}
        if e.op == "concat":
            return l.str.concat(r)
        return ops[e.op](r)
    if isinstance(e, P.UnaryOp):
        c = expr_to_daft(e.child, binder)
        return ~c if e.op == "not" else (lit(0) - c)
    if isinstance(e, P.BetweenExpr):
        out = expr_to_daft(e.child, binder).between(
            expr_to_daft(e.lo, binder), expr_to_daft(e.hi, binder))
        return ~out if e.negated else out
    if isinstance(e, P.InList):
        vals = []
        for v in e.values:
            f = _fold_date(v)
            if f is not None:
                vals.append(f)
            elif isinstance(v, P.Lit):
                vals.append(v.value)
            else:
                raise SQLPlanError("IN list values must be literals")
        out = expr_to_daft(e.child, binder).is_in(vals)
        return ~out if e.negated else out
    if isinstance(e, P.LikeExpr):
        c = expr_to_daft(e.child, binder)
        out = c.str.ilike(e.pattern) if e.case_insensitive \
            else c.str.like(e.pattern)
        return ~out if e.negated else out
    if isinstance(e, P.IsNullExpr):
        c = expr_to_daft(e.child, binder)
        return c.not_null() if e.negated else c.is_null()
    if isinstance(e, P.CaseExpr):
        default = expr_to_daft(e.default, binder) if e.default is not None \
            else lit(None)
        out = default
        for cond, val in reversed(e.whens):
            out = expr_to_daft(cond, binder).if_else(
                expr_to_daft(val, binder), out)
        return out
    if isinstance(e, P.CastExpr):
        c = expr_to_daft(e.child, binder)
        tn = e.type_name
        m = {"int": DataType.int64(), "integer": DataType.int64(),
             "bigint": DataType.int64(), "smallint": DataType.int16(),
             "double": DataType.float64(), "double precision":
             DataType.float64(), "float": DataType.float64(),
             "real": DataType.float32(), "decimal": DataType.float64(),
             "numeric": DataType.float64(), "varchar": DataType.string(),
             "text": DataType.string(), "char": DataType.string(),
             "date": DataType.date(), "boolean": DataType.bool()}
        import re as _re
        dm = _re.fullmatch(r"(?:decimal|numeric)\s*\((\d+)\s*,\s*(\d+)\)",
                           tn)
        if dm:
            return c.cast(DataType.decimal128(int(dm.group(1)),
                                              int(dm.group(2))))
        base = tn.split("(")[0].strip()
        if tn not in m and base in m:
            tn = base  # length-parameterized types: varchar(n), char(n)...
        if tn not in m:
            raise SQLPlanError(f"unsupported cast type {tn}")
        return c.cast(m[tn])
    if isinstance(e, P.ExtractExpr):
        c = expr_to_daft(e.child, binder)
        ns = c.dt
        fns = {"year": ns.year, "month": ns.month, "day": ns.day,
               "quarter": ns.quarter, "hour": ns.hour, "minute": ns.minute,
               "second": ns.second}
        if e.part not in fns:
            raise SQLPlanError(f"unsupported EXTRACT part {e.part}")
        return fns[e.part]()
    if isinstance(e, P.SubstringExpr):
        c = expr_to_daft(e.child, binder)
        start = e.start.value if isinstance(e.start, P.Lit) else None
        if start is None:
            raise SQLPlanError("SUBSTRING start must be a literal")
        length = None
        if e.length is not None:
            if not isinstance(e.length, P.Lit):
                raise SQLPlanError("SUBSTRING length must be a literal")
            length = int(e.length.value)
        return c.str.substr(int(start) - 1, length)
    if isinstance(e, P.FuncCall):
        return _bind_func(e, binder)
    if isinstance(e, (P.SubqueryExpr, P.InSubquery, P.ExistsExpr)):
        raise SQLPlanError(
            "subqueries are only supported as top-level WHERE/HAVING "
            "conjuncts")
    raise SQLPlanError(f"cannot bind SQL expression {e!r}")


def _bind_func(e: P.FuncCall, binder) -> Expression:
    name = e.name
    if name in _AGG_FUNCS:
        if e.star or (name == "count" and not e.args):
            return Expression(Agg(AggKind.COUNT_ALL, None))
        child = expr_to_daft(e.args[0], binder)
        if e.distinct:
            if name == "count":
                return child.count_distinct()
            raise SQLPlanError(f"DISTINCT not supported for {name}")
        return Expression(Agg(_AGG_FUNCS[name], child._node))
    args = [expr_to_daft(a, binder) for a in e.args]
    simple = {
        "abs": lambda a: a.abs(), "round": lambda a, *r: a.round(
            int(r[0]._node.value) if r else 0),
        "floor": lambda a: a.floor(), "ceil": lambda a: a.ceil(),
        "ceiling": lambda a: a.ceil(), "sqrt": lambda a: a.sqrt(),
        "exp": lambda a: a.exp(), "ln": lambda a: a.log(),
        "lower": lambda a: a.str.lower(), "upper": lambda a: a.str.upper(),
        "length": lambda a: a.str.length(),
        "char_length": lambda a: a.str.length(),
        "trim": lambda a: a.str.strip(),
        "ltrim": lambda a: a.str.lstrip(), "rtrim": lambda a: a.str.rstrip(),
        "contains": lambda a, b: a.str.contains(_litval(b)),
        "starts_with": lambda a, b: a.str.startswith(_litval(b)),
        "ends_with": lambda a, b: a.str.endswith(_litval(b)),
        "year": lambda a: a.dt.year, "month": lambda a: a.dt.month,
        "day": lambda a: a.dt.day,
    }
    if name in ("year", "month", "day"):
        return simple[name](args[0])()
    if name == "coalesce":
        from ..functions import coalesce
        return coalesce(*args)
    if name == "substr" or name == "substring":
        start = int(_litval_num(e.args[1], binder)) - 1
        length = int(_litval_num(e.args[2], binder)) if len(e.args) > 2 \
            else None
        return args[0].str.substr(start, length)
    if name == "concat":
        out = args[0]
        for a in args[1:]:
            out = out.str.concat(a)
        return out
    if name == "nullif":
        a, b = args
        return (a == b).if_else(lit(None), a)
    if name in ("greatest", "least"):
        out = args[0]
        for a in args[1:]:
            cmp = (a > out) if name == "greatest" else (a < out)
            out = cmp.if_else(a, out)
        return out
    if name in ("if", "iff"):
        return args[0].if_else(args[1], args[2])
    if name == "ifnull" or name == "nvl":
        return args[0].fill_null(args[1])
    if name in simple:
        return simple[name](*args)
    # fall back to the free-function registry (daft.functions parity
    # surface: sin/levenshtein_distance/to_snake_case/...)
    from .. import functions as F
    fn = getattr(F, name, None)
    if fn is not None and callable(fn):
        try:
            return fn(*args)
        except TypeError as te:
            raise SQLPlanError(f"bad arguments for {name}(): {te}")
    raise SQLPlanError(f"unknown function {name}")


def _litval(e: Expression):
    from ..expressions.expressions import Literal
    assert isinstance(e._node, Literal)
    return e._node.value


def _litval_num(a, binder):
    if isinstance(a, P.Lit):
        return a.value
    raise SQLPlanError("expected literal argument")


# ---------------------------------------------------------------------------
# conjunct utilities
# ---------------------------------------------------------------------------

def _split_conj(e) -> List[Any]:
    if isinstance(e, P.BinOp) and e.op == "and":
        return _split_conj(e.left) + _split_conj(e.right)
    return [e]


def _split_disj(e) -> List[Any]:
    if isinstance(e, P.BinOp) and e.op == "or":
        return _split_disj(e.left) + _split_disj(e.right)
    return [e]


def _hoist_common_from_or(conjs: List[Any]) -> List[Any]:
    """`(a and X) or (a and Y)` implies `a`: hoist equality conjuncts common
    to every OR branch so they are available for equi-join extraction
    (ref shape: TPC-H Q19)."""
    out = list(conjs)
    for cj in conjs:
        branches = _split_disj(cj)
        if len(branches) < 2:
            continue
        branch_sets = [_split_conj(b) for b in branches]
        for cand in branch_sets[0]:
            if not (isinstance(cand, P.BinOp) and cand.op == "eq"):
                continue
            if all(any(cand == x for x in bs) for bs in branch_sets[1:]):
                if not any(cand == x for x in out):
                    out.append(cand)
    return out


def _has_subquery(e) -> bool:
    if isinstance(e, (P.SubqueryExpr, P.InSubquery, P.ExistsExpr)):
        return True
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, (list, tuple)):
            if any(_has_subquery(x) for x in v
                   if hasattr(x, "__dataclass_fields__")):
                return True
        elif hasattr(v, "__dataclass_fields__") and _has_subquery(v):
            return True
    return False


def _col_refs(e, out: List[P.Col]):
    if isinstance(e, P.Col):
        out.append(e)
        return
    if isinstance(e, (P.SubqueryExpr, P.InSubquery, P.ExistsExpr)):
        return  # inner scope
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, (list, tuple)):
            for x in v:
                if isinstance(x, tuple):
                    for y in x:
                        if hasattr(y, "__dataclass_fields__"):
                            _col_refs(y, out)
                elif hasattr(x, "__dataclass_fields__"):
                    _col_refs(x, out)
        elif hasattr(v, "__dataclass_fields__"):
            _col_refs(v, out)


# ---------------------------------------------------------------------------
# FROM planning with equi-join extraction
# ---------------------------------------------------------------------------

def _plan_values(t: P.TableRef):
    """(VALUES (..), (..)) AS v(c1, c2): evaluate each row's constant
    expressions over a one-row dummy frame (ref: daft-sql VALUES)."""
    import daft_amd as daft
    dummy = daft.from_pydict({"__one__": [0]})
    binder = Binder({"__v__": {}})
    ncols = len(t.values[0])
    cols = {}
    for ci in range(ncols):
        vals = []
        for row in t.values:
            if len(row) != ncols:
                raise SQLPlanError("VALUES rows have differing arity")
            e = expr_to_daft(row[ci], binder)
            vals.append(dummy.select(e.alias("x")).to_pydict()["x"][0])
        cols[f"column{ci + 1}"] = vals
    return daft.from_pydict(cols)


def _plan_table_fn(t: P.TableRef):
    """FROM-position table functions: read_parquet/read_csv/read_json/
    read_ipc('path', ...) (ref: daft-sql table providers)."""
    import daft_amd as daft
    fns = {"read_parquet": daft.read_parquet, "read_csv": daft.read_csv,
           "read_json": daft.read_json,
           "read_ipc": getattr(daft, "read_ipc", None)}
    fn = fns.get(t.fn)
    if fn is None:
        raise SQLPlanError(f"unknown table function {t.fn!r}")
    dummy_binder = Binder({"__v__": {}})
    args = []
    for a in t.fn_args or []:
        if isinstance(a, P.Lit):
            args.append(a.value)
        else:
            import daft_amd as daft
            d = daft.from_pydict({"__one__": [0]})
            e = expr_to_daft(a, dummy_binder)
            args.append(d.select(e.alias("x")).to_pydict()["x"][0])
    return fn(*args)


class _FromPlanner:
    def __init__(self, lookup, outer: Optional[Binder]):
        self.lookup = lookup
        self.outer = outer
        self.binder = Binder({}, outer=outer)
        self.df = None
        self.used_names: Set[str] = set()

    def _plan_table(self, t: P.TableRef, ctes):
        alias = t.alias or t.name
        if t.subquery is not None:
            df = _plan_select(t.subquery, self.lookup, outer=self.outer,
                              ctes=ctes)
        elif t.values is not None:
            df = _plan_values(t)
        elif t.fn is not None:
            df = _plan_table_fn(t)
        elif t.name in ctes:
            df = ctes[t.name]
        else:
            df = self.lookup(t.name)
        if t.col_names:
            old = df.column_names()
            if len(t.col_names) != len(old):
                raise SQLPlanError(
                    f"alias column list has {len(t.col_names)} names for "
                    f"{len(old)} columns")
            df = df.with_columns_renamed(dict(zip(old, t.col_names)))
        mapping = {}
        renames = {}
        for c in df.column_names():
            out = c
            if out in self.used_names:
                out = f"{alias}__{c}"
                i = 2
                while out in self.used_names:
                    out = f"{alias}{i}__{c}"
                    i += 1
                renames[c] = out
            mapping[c] = out
            self.used_names.add(out)
        if renames:
            df = df.with_columns_renamed(renames)
        return df, alias, mapping

    def add_first(self, t: P.TableRef, ctes):
        df, alias, mapping = self._plan_table(t, ctes)
        self.df = df
        self.binder.tables[alias] = mapping

    def add_joined(self, t: P.TableRef, how: str, on_conjs: List[Any],
                   ctes):
        """Join table t using the equality conjuncts that connect it to the
        current set; returns (conjuncts consumed, right-side filters)."""
        df, alias, mapping = self._plan_table(t, ctes)
        trial = Binder(dict(self.binder.tables), outer=self.outer)
        trial.tables[alias] = mapping
        new_binder = Binder({alias: mapping}, outer=self.outer)
        left_on, right_on, used, right_filters = [], [], [], []
        for cj in on_conjs:
            side = self._equi_sides(cj, alias, trial)
            if side is not None:
                le, re_ = side
                left_on.append(le)
                right_on.append(re_)
                used.append(cj)
                continue
            # conjunct referencing only the new table: pre-filter the right
            # side (required for outer-join ON semantics, ref Q13)
            refs = []
            _col_refs(cj, refs)
            if refs and all(self._in_table(r, alias, trial) for r in refs) \
                    and not _has_subquery(cj):
                df = df.where(expr_to_daft(cj, new_binder))
                right_filters.append(cj)
        if not left_on and how == "inner":
            how = "cross"
        if how == "cross":
            self.df = self.df.join(df, how="cross")
        else:
            self.df = self.df.join(df, left_on=left_on, right_on=right_on,
                                   how=how, suffix="__r")
        self.binder.tables[alias] = mapping
        return used, right_filters

    def _equi_sides(self, cj, new_alias: str, trial: Binder):
        """If cj is `a = b` with one side fully in the current tables and the
        other fully in new_alias, return bound (left_expr, right_expr)."""
        if not (isinstance(cj, P.BinOp) and cj.op == "eq"):
            return None
        refs_l: List[P.Col] = []
        refs_r: List[P.Col] = []
        _col_refs(cj.left, refs_l)
        _col_refs(cj.right, refs_r)
        if not refs_l or not refs_r:
            return None

        def side_of(refs):
            in_new = all(self._in_table(r, new_alias, trial) for r in refs)
            in_cur = all(self._in_current(r) for r in refs)
            if in_new and not in_cur:
                return "new"
            if in_cur:
                return "cur"
            return None
        sl, sr = side_of(refs_l), side_of(refs_r)
        new_binder = Binder({new_alias: trial.tables[new_alias]},
                            outer=self.outer)
        if sl == "cur" and sr == "new":
            return (expr_to_daft(cj.left, self.binder),
                    expr_to_daft(cj.right, new_binder))
        if sl == "new" and sr == "cur":
            return (expr_to_daft(cj.right, self.binder),
                    expr_to_daft(cj.left, new_binder))
        return None

    def _probe_cols(self, t: P.TableRef, ctes) -> Optional[Set[str]]:
        if t.subquery is not None:
            return None
        if t.name in ctes:
            return set(ctes[t.name].column_names())
        try:
            return set(self.lookup(t.name).column_names())
        except Exception:
            return None

    def connects(self, t: P.TableRef, conjs: List[Any], ctes) -> bool:
        """Does any equality conjunct link table t to the current set?
        Used to order comma-joined FROM tables by connectivity so that
        `FROM part, supplier, lineitem, ...` never plans a cross join
        (the TPC-H q2/q8/q9 shape; ref: rules/eliminate_cross_join.rs)."""
        cols = self._probe_cols(t, ctes)
        if cols is None:
            return False
        alias = t.alias or t.name

        def in_cand(r: P.Col) -> bool:
            if r.table is not None:
                return r.table == alias and r.name in cols
            return r.name in cols

        for cj in conjs:
            if not (isinstance(cj, P.BinOp) and cj.op == "eq"):
                continue
            rl: List[P.Col] = []
            rr: List[P.Col] = []
            _col_refs(cj.left, rl)
            _col_refs(cj.right, rr)
            if not rl or not rr:
                continue
            l_cand = all(in_cand(r) for r in rl)
            r_cand = all(in_cand(r) for r in rr)
            l_cur = all(self._in_current(r) for r in rl)
            r_cur = all(self._in_current(r) for r in rr)
            if (l_cand and r_cur and not l_cur) or \
                    (r_cand and l_cur and not r_cur):
                return True
        return False

    def _in_table(self, r: P.Col, alias: str, trial: Binder) -> bool:
        m = trial.tables.get(alias, {})
        if r.table is not None:
            return r.table == alias and r.name in m
        return r.name in m and self.binder.resolve(None, r.name) is None

    def _in_current(self, r: P.Col) -> bool:
        try:
            return self.binder.resolve(r.table, r.name) is not None
        except SQLPlanError:
            return True  # ambiguous -> definitely current


# ---------------------------------------------------------------------------
# SELECT planning
# ---------------------------------------------------------------------------

def _plan_select(stmt: P.SelectStmt, lookup, outer: Optional[Binder],
                 ctes: Optional[dict] = None):
    from ..dataframe import DataFrame
    ctes = dict(ctes or {})
    for name, sub in stmt.ctes:
        ctes[name] = _plan_select(sub, lookup, outer=None, ctes=ctes)

    fp = _FromPlanner(lookup, outer)
    if not stmt.from_tables:
        # constant projection: one dummy row (ref: daft-sql SELECT 1+1)
        import daft_amd as daft
        fp.df = daft.from_pydict({"__one__": [0]})
        fp.binder.tables["__dual__"] = {}
    else:
        fp.add_first(stmt.from_tables[0], ctes)

    where_conjs = _split_conj(stmt.where) if stmt.where is not None else []
    where_conjs = _hoist_common_from_or(where_conjs)
    plain = [c for c in where_conjs if not _has_subquery(c)]
    subq = [c for c in where_conjs if _has_subquery(c)]

    # comma tables: connect via extracted equality conjuncts, picking the
    # next table by connectivity (never cross-join when an equi edge exists)
    remaining = list(stmt.from_tables[1:])
    while remaining:
        pick = next((t for t in remaining if fp.connects(t, plain, ctes)),
                    remaining[0])
        remaining.remove(pick)
        used, right_filters = fp.add_joined(pick, "inner", plain, ctes)
        for u in used:
            plain.remove(u)
        for rf in right_filters:   # already applied to the joined side
            if rf in plain:
                plain.remove(rf)
    for jc in stmt.joins:
        on_conjs = _split_conj(jc.on) if jc.on is not None else []
        used, right_filters = fp.add_joined(jc.table, jc.how, on_conjs, ctes)
        leftover = [c for c in on_conjs
                    if c not in used and c not in right_filters]
        if leftover:
            if jc.how not in ("inner", "cross"):
                raise SQLPlanError("non-equi ON conditions only supported "
                                   "for inner joins")
            plain.extend(leftover)

    df = fp.df
    binder = fp.binder

    # plain predicates
    for cj in plain:
        df = df.where(expr_to_daft(cj, binder))

    # subquery predicates (top-level conjuncts only); a pair of
    # EXISTS/NOT-EXISTS over the same correlated group fuses into ONE
    # groupby + left join (the q21 shape)
    fused = _try_fuse_exists_pair(df, subq, binder, lookup, ctes)
    if fused is not None:
        df, consumed = fused
        subq = [c for c in subq if not any(c is k for k in consumed)]
    for cj in subq:
        df = _apply_subquery_conjunct(df, cj, binder, lookup, ctes)

    # scalar subqueries in the SELECT list (ref: planner.rs
    # SQLExpr::Subquery): uncorrelated evaluate to literals; equality-
    # correlated decorrelate into a grouped LEFT join (missing groups
    # yield NULL, matching scalar-subquery semantics)
    if any((not it.star) and _has_subquery(it.expr) for it in stmt.items):
        df, stmt = _rewrite_select_subqueries(df, stmt, binder, lookup,
                                              ctes)

    # GROUP BY ALL: every non-aggregate select item is a group key
    if any(g == "__GROUP_BY_ALL__" for g in stmt.group_by):
        import dataclasses
        gb = [it.expr for it in stmt.items
              if not it.star and not _has_aggregate(it.expr)]
        stmt = dataclasses.replace(stmt, group_by=gb)

    # aggregate or plain projection
    has_agg = any(_has_aggregate(it.expr) for it in stmt.items
                  if not it.star) or stmt.group_by or \
        (stmt.having is not None)

    select_aliases: Dict[str, Any] = {}
    for it in stmt.items:
        if it.alias and not it.star:
            select_aliases[it.alias] = it.expr

    # window functions: compute as extra columns before projection
    win_map = {}
    _wsub_all: List[Tuple[str, Any]] = []
    agg_win = has_agg and any(_has_window(it.expr) for it in stmt.items)
    if not agg_win:
        for i, it in enumerate(stmt.items):
            if isinstance(it.expr, P.WindowExpr):
                wname = it.alias or it.expr.func.name
                df = df.with_window_columns(
                    {wname: _window_to_daft(it.expr, binder)})
                win_map[i] = wname
            elif _has_window(it.expr):
                # window embedded in an expression: hidden window
                # columns + residual evaluated afterwards (shared name
                # counter keeps hidden columns unique across items)
                wsub = _wsub_all
                n0 = len(wsub)
                residual = _extract_window_subtrees(it.expr, wsub)
                for nm, we in wsub[n0:]:
                    df = df.with_window_columns(
                        {nm: _window_to_daft(we, binder)})
                binder.tables["__win__"] = {nm: nm for nm, _ in wsub}
                name = it.alias or _default_name(it.expr, binder)
                df = df.with_column(name,
                                    expr_to_daft(residual, binder))
                del binder.tables["__win__"]
                win_map[i] = name

    if agg_win:
        # windows over GROUP BY results (rank() OVER (ORDER BY sum(v))):
        # aggregate first, then window over the aggregated frame with
        # aggregate sub-expressions rewritten to their output columns
        # (hidden extra aggregates when the window uses one the SELECT
        # list doesn't) — ref: daft-sql window-over-aggregate planning
        import dataclasses as _dc
        name_of = []
        for it in stmt.items:
            if it.star or _has_window(it.expr):
                continue
            name_of.append((it.expr, it.alias or
                            _default_name(it.expr, binder)))
        hidden: List[Tuple[Any, str]] = []

        def _rw(node):
            for ast, nm in name_of:
                if node == ast:
                    return P.Col(None, nm)
            if isinstance(node, P.WindowExpr):
                # the window FUNC head itself is not a group aggregate —
                # rewrite only its arguments / partition / order exprs
                fn = node.func
                new_fn = P.FuncCall(fn.name, [_rw(a) for a in fn.args],
                                    fn.distinct, fn.star) \
                    if isinstance(fn, P.FuncCall) else _rw(fn)
                return P.WindowExpr(
                    new_fn, [_rw(x) for x in node.partition_by],
                    [(_rw(e), d) for e, d in node.order_by], node.frame)
            if isinstance(node, P.FuncCall) and _has_aggregate(node):
                for h_ast, h_nm in hidden:
                    if node == h_ast:
                        return P.Col(None, h_nm)
                h_nm = f"__wagg{len(hidden)}"
                hidden.append((node, h_nm))
                return P.Col(None, h_nm)
            if _dc.is_dataclass(node) and not isinstance(node, type):
                kw = {}
                for f in _dc.fields(node):
                    v = getattr(node, f.name)
                    if isinstance(v, list):
                        kw[f.name] = [
                            tuple(_rw(y) if _dc.is_dataclass(y) else y
                                  for y in x) if isinstance(x, tuple)
                            else (_rw(x) if _dc.is_dataclass(x) else x)
                            for x in v]
                    elif _dc.is_dataclass(v) and not isinstance(v, type):
                        kw[f.name] = _rw(v)
                    else:
                        kw[f.name] = v
                return type(node)(**kw)
            return node

        rewritten = {}
        for i, it in enumerate(stmt.items):
            if _has_window(it.expr):
                rewritten[i] = _rw(it.expr)
        base_items = [it for it in stmt.items
                      if not _has_window(it.expr)]
        base_items += [P.SelectItem(ast, nm) for ast, nm in hidden]
        agg_stmt = _dc.replace(stmt, items=base_items, order_by=[],
                               limit=None, offset=None, qualify=None,
                               distinct=False)
        df = _plan_aggregate(df, agg_stmt, binder, select_aliases,
                             lookup, ctes)
        post_binder = Binder(
            {"__agg__": {c: c for c in df.column_names()}})
        for i, it in enumerate(stmt.items):
            if i not in rewritten:
                continue
            wname = it.alias or _default_name(it.expr, binder) \
                if not isinstance(it.expr, P.WindowExpr) \
                else (it.alias or it.expr.func.name)
            wsub = _wsub_all
            n0 = len(wsub)
            residual = _extract_window_subtrees(rewritten[i], wsub)
            for nm, we in wsub[n0:]:
                df = df.with_window_columns(
                    {nm: _window_to_daft(we, post_binder)})
            pb2 = Binder({"__agg__": {c: c for c in df.column_names()}})
            df = df.with_column(wname, expr_to_daft(residual, pb2))
            win_map[i] = wname
        sel = []
        for i, it in enumerate(stmt.items):
            if i in win_map:
                sel.append(col(win_map[i]))
            else:
                sel.append(col(it.alias or
                               _default_name(it.expr, binder)))
        df = df.select(*sel)
        out_names = df.column_names()
    elif has_agg:
        df = _plan_aggregate(df, stmt, binder, select_aliases, lookup,
                             ctes)
        out_names = _output_names(stmt, binder)
    else:
        exprs = []
        for i, it in enumerate(stmt.items):
            if it.star:
                exprs.extend(col(n) for n in df.column_names()
                             if not n.startswith(("__ssv", "__ssk")))
            elif i in win_map:
                exprs.append(col(win_map[i]))
            else:
                e = expr_to_daft(it.expr, binder)
                name = it.alias or _default_name(it.expr, binder)
                exprs.append(e.alias(name))
        df = df.select(*exprs)
        out_names = df.column_names()

    if stmt.qualify is not None:
        # QUALIFY filters on window results after projection; inline
        # window exprs compute as hidden columns first
        q = stmt.qualify
        hidden = []

        def _rewrite(e):
            if isinstance(e, P.WindowExpr):
                hname = f"__q{len(hidden)}"
                hidden.append((hname, e))
                return P.Col(None, hname)
            import dataclasses as _dc
            if _dc.is_dataclass(e):
                kw = {}
                for f in _dc.fields(e):
                    v = getattr(e, f.name)
                    if isinstance(v, list):
                        kw[f.name] = [_rewrite(x) if _dc.is_dataclass(x)
                                      or isinstance(x, P.WindowExpr)
                                      else x for x in v]
                    elif _dc.is_dataclass(v) or isinstance(v, P.WindowExpr):
                        kw[f.name] = _rewrite(v)
                    else:
                        kw[f.name] = v
                return type(e)(**kw)
            return e

        q = _rewrite(q)
        for hname, we in hidden:
            df = df.with_window_columns(
                {hname: _window_to_daft(we, binder)})
        pred = _bind_order_expr(q, out_names, select_aliases, binder, stmt)
        df = df.where(pred)
        if hidden:
            df = df.select(*[col(n) for n in out_names])

    if stmt.distinct:
        df = df.distinct()

    if stmt.order_by:
        by, desc, nf = [], [], []
        for ob in stmt.order_by:
            by.append(_bind_order_expr(ob.expr, out_names, select_aliases,
                                       binder, stmt))
            desc.append(ob.desc)
            nf.append(ob.nulls_first)
        df = df.sort(by, desc=desc,
                     nulls_first=[d if f is None else f
                                  for f, d in zip(nf, desc)])
    if stmt.limit is not None:
        if stmt.offset:
            df = df.offset(stmt.offset)
        df = df.limit(stmt.limit)
    return df


def _window_to_daft(we, binder):
    """WindowExpr AST -> Expression bound to a Window spec (ref:
    daft-sql window planning)."""
    from ..window import Window
    from .. import functions as F
    w = Window()
    if we.partition_by:
        w = w.partition_by(*[expr_to_daft(p, binder)
                             for p in we.partition_by])
    if we.order_by:
        w = w.order_by(*[expr_to_daft(e, binder) for e, _ in we.order_by],
                       desc=[d for _, d in we.order_by])
    if we.frame is not None:
        lo, hi = we.frame
        lo = Window.unbounded_preceding if lo is None else lo
        hi = Window.unbounded_following if hi is None else hi
        w = w.rows_between(lo, hi)
    fn = we.func.name
    args = we.func.args
    if fn in ("row_number", "rank", "dense_rank"):
        return getattr(F, fn)().over(w)
    if fn in ("first_value", "last_value"):
        e = expr_to_daft(args[0], binder)
        make = F.w_first_value if fn == "first_value" else F.w_last_value
        return make(e).over(w)
    if fn in ("lag", "lead"):
        e = expr_to_daft(args[0], binder)
        off = args[1].value if len(args) > 1 else 1
        dflt = args[2].value if len(args) > 2 else None
        return getattr(e, fn)(int(off), dflt).over(w)
    if fn in ("sum", "avg", "min", "max", "count", "mean", "stddev"):
        if we.func.star:
            from ..expressions.expressions import Agg, AggKind, Expression
            return Expression(Agg(AggKind.COUNT_ALL, None)).over(w)
        e = expr_to_daft(args[0], binder)
        agg = {"sum": e.sum, "avg": e.mean, "mean": e.mean, "min": e.min,
               "max": e.max, "count": e.count, "stddev": e.stddev}[fn]()
        return agg.over(w)
    raise SQLPlanError(f"unsupported window function {fn!r}")



def _has_window(e) -> bool:
    if isinstance(e, P.WindowExpr):
        return True
    import dataclasses as _dc
    if _dc.is_dataclass(e) and not isinstance(e, type):
        for f in _dc.fields(e):
            v = getattr(e, f.name)
            if isinstance(v, (list, tuple)):
                for x in v:
                    if isinstance(x, tuple):
                        if any(_has_window(y) for y in x):
                            return True
                    elif _has_window(x):
                        return True
            elif _has_window(v):
                return True
    return False


def _extract_window_subtrees(ast, out):
    """Replace every WindowExpr subtree with Col(__w{i}); append
    (name, window_ast) to `out`.  Returns the residual AST."""
    import dataclasses as _dc
    if isinstance(ast, P.WindowExpr):
        nm = f"__w{len(out)}"
        out.append((nm, ast))
        return P.Col(None, nm)
    if _dc.is_dataclass(ast) and not isinstance(ast, type):
        kw = {}
        for f in _dc.fields(ast):
            v = getattr(ast, f.name)
            if isinstance(v, list):
                kw[f.name] = [
                    tuple(_extract_window_subtrees(y, out)
                          if _dc.is_dataclass(y) else y for y in x)
                    if isinstance(x, tuple) else
                    (_extract_window_subtrees(x, out)
                     if _dc.is_dataclass(x) else x) for x in v]
            elif _dc.is_dataclass(v) and not isinstance(v, type):
                kw[f.name] = _extract_window_subtrees(v, out)
            else:
                kw[f.name] = v
        return type(ast)(**kw)
    return ast


def _has_aggregate(e) -> bool:
    if e is None:
        return False
    if isinstance(e, P.WindowExpr):
        return False  # window aggregates are not GROUP BY aggregates
    if isinstance(e, P.FuncCall) and (e.name in _AGG_FUNCS or e.star):
        return True
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if isinstance(v, (list, tuple)):
            for x in v:
                if isinstance(x, tuple):
                    if any(_has_aggregate(y) for y in x
                           if hasattr(y, "__dataclass_fields__")):
                        return True
                elif hasattr(x, "__dataclass_fields__") and _has_aggregate(x):
                    return True
        elif hasattr(v, "__dataclass_fields__") and _has_aggregate(v):
            return True
    return False


def _default_name(e, binder) -> str:
    if isinstance(e, P.Col):
        return e.name
    try:
        return expr_to_daft(e, binder).name()
    except Exception:
        return "col"


def _output_names(stmt, binder) -> List[str]:
    out = []
    for it in stmt.items:
        if it.star:
            continue
        out.append(it.alias or _default_name(it.expr, binder))
    return out


def _bind_order_expr(e, out_names: List[str], select_aliases, binder, stmt):
    if isinstance(e, P.Lit) and isinstance(e.value, int):
        return col(out_names[e.value - 1])
    if isinstance(e, P.Col) and e.table is None and e.name in out_names:
        return col(e.name)
    if isinstance(e, P.Col) and e.table is None and e.name in select_aliases:
        return col(e.name)
    # expression over output columns (bind against output names directly)
    try:
        return expr_to_daft(e, None)
    except Exception:
        return expr_to_daft(e, binder)


def _plan_aggregate(df, stmt: P.SelectStmt, binder: Binder,
                    select_aliases: Dict[str, Any], lookup=None,
                    ctes=None):
    # resolve group keys: alias -> select expr; ordinal -> select item; expr
    group_exprs = []
    group_key_asts = []
    for g in stmt.group_by:
        if isinstance(g, P.Lit) and isinstance(g.value, int):
            it = stmt.items[g.value - 1]
            ast = it.expr
            name = it.alias or _default_name(ast, binder)
        elif isinstance(g, P.Col) and g.table is None and \
                g.name in select_aliases and \
                binder.resolve(None, g.name) is None:
            ast = select_aliases[g.name]
            name = g.name
        else:
            ast = g
            name = _default_name(g, binder)
        group_exprs.append(expr_to_daft(ast, binder).alias(name))
        group_key_asts.append((ast, name))

    def rewrite_item(e):
        """Replace group-key sub-expressions with their output columns, bind
        aggregates."""
        for ast, name in group_key_asts:
            if e == ast or (isinstance(e, P.Col) and e.table is None and
                            e.name == name and
                            binder.resolve(None, e.name) is None):
                return col(name)
        return None

    agg_exprs = []
    finals = []
    for it in stmt.items:
        if it.star:
            raise SQLPlanError("SELECT * with GROUP BY is not supported")
        name = it.alias or _default_name(it.expr, binder)
        hit = rewrite_item(it.expr)
        if hit is not None:
            # pure group key (possibly under a different output alias)
            finals.append(hit.alias(name))
            continue
        if isinstance(it.expr, P.Col) and binder.resolve(
                it.expr.table, it.expr.name) is not None and \
                not _has_aggregate(it.expr):
            # bare column that is functionally dependent on the group keys
            agg_exprs.append(expr_to_daft(it.expr, binder).any_value()
                             .alias(name))
        else:
            agg_exprs.append(expr_to_daft(it.expr, binder).alias(name))
        finals.append(col(name))

    having_expr = None
    if stmt.having is not None:
        if _has_subquery(stmt.having):
            # uncorrelated scalar subqueries inside HAVING: inline literal
            having_ast = _inline_uncorrelated(stmt.having, binder, lookup,
                                              ctes)
        else:
            having_ast = stmt.having
        having_expr = expr_to_daft(having_ast, binder)
        hname = "__having"
        agg_exprs.append(having_expr.alias(hname))
        having_expr = col(hname)

    grouped = df._builder.aggregate(agg_exprs, group_exprs)
    from ..dataframe import DataFrame
    out = DataFrame(grouped)
    if having_expr is not None:
        out = out.where(having_expr)
    return out.select(*finals)


def _rewrite_select_subqueries(df, stmt, binder, lookup, ctes):
    import dataclasses as _dc
    counter = [0]
    dfbox = [df]

    def walk(node):
        if isinstance(node, P.SubqueryExpr):
            corr = _correlation_info(node.query, binder, lookup, ctes)
            if corr is None:
                return P.Lit(_eval_scalar_subquery(node.query, binder,
                                                   lookup, ctes))
            outer_cols, inner_cols, residual = corr
            sub = node.query
            if len(sub.items) != 1 or sub.items[0].star:
                raise SQLPlanError(
                    "scalar subquery must select one column")
            agg_ast = sub.items[0].expr
            if not _has_aggregate(agg_ast):
                # a well-formed scalar subquery yields <= 1 row per
                # outer key; MAX selects that single value (multi-row
                # subqueries are rejected at runtime by other engines)
                agg_ast = P.FuncCall("max", [agg_ast])
            n = counter[0]
            counter[0] += 1
            vname = f"__ssv{n}"
            keys = [f"__ssk{n}_{i}" for i in range(len(inner_cols))]
            inner_stmt = P.SelectStmt(
                items=[P.SelectItem(c, k)
                       for c, k in zip(inner_cols, keys)] +
                      [P.SelectItem(agg_ast, vname)],
                from_tables=sub.from_tables, joins=sub.joins,
                where=_rebuild_where(residual),
                group_by=list(inner_cols))
            sub_df = _plan_select(inner_stmt, lookup, outer=None,
                                  ctes=ctes)
            left = [expr_to_daft(c, binder) for c in outer_cols]
            dfbox[0] = dfbox[0].join(sub_df, left_on=left,
                                     right_on=[col(k) for k in keys],
                                     how="left")
            binder.tables.setdefault("__ssv__", {})[vname] = vname
            return P.Col(None, vname)
        if _dc.is_dataclass(node) and not isinstance(node, type):
            kw = {}
            for f in _dc.fields(node):
                v = getattr(node, f.name)
                if isinstance(v, list):
                    kw[f.name] = [
                        tuple(walk(y) if _dc.is_dataclass(y) else y
                              for y in x) if isinstance(x, tuple)
                        else (walk(x) if _dc.is_dataclass(x) else x)
                        for x in v]
                elif _dc.is_dataclass(v) and not isinstance(v, type):
                    kw[f.name] = walk(v)
                else:
                    kw[f.name] = v
            return type(node)(**kw)
        return node

    new_items = []
    for it in stmt.items:
        if it.star or not _has_subquery(it.expr):
            new_items.append(it)
        else:
            new_items.append(P.SelectItem(walk(it.expr), it.alias))
    stmt = _dc.replace(stmt, items=new_items)
    return dfbox[0], stmt


def _inline_uncorrelated(e, binder, lookup, ctes):
    """Replace uncorrelated scalar subqueries inside an expression AST with
    literal values (used for HAVING thresholds, ref Q11)."""
    if isinstance(e, P.SubqueryExpr):
        val = _eval_scalar_subquery(e.query, binder, lookup, ctes)
        return P.Lit(val)
    for f in getattr(e, "__dataclass_fields__", {}):
        v = getattr(e, f)
        if hasattr(v, "__dataclass_fields__"):
            setattr(e, f, _inline_uncorrelated(v, binder, lookup, ctes))
        elif isinstance(v, list):
            setattr(e, f, [
                _inline_uncorrelated(x, binder, lookup, ctes)
                if hasattr(x, "__dataclass_fields__") else x for x in v])
    return e


def _eval_scalar_subquery(sub: P.SelectStmt, binder, lookup, ctes) -> Any:
    df = _plan_select(sub, lookup, outer=binder, ctes=ctes)
    d = df.to_pydict()
    colname = next(iter(d))
    vals = d[colname]
    return vals[0] if vals else None


# ---------------------------------------------------------------------------
# subquery conjuncts
# ---------------------------------------------------------------------------

def _apply_subquery_conjunct(df, cj, binder: Binder, lookup, ctes):
    if isinstance(cj, P.ExistsExpr) or (
            isinstance(cj, P.UnaryOp) and cj.op == "not" and
            isinstance(cj.child, P.ExistsExpr)):
        negated = isinstance(cj, P.UnaryOp)
        ex = cj.child if negated else cj
        negated = negated or ex.negated
        return _plan_exists(df, ex.query, negated, binder, lookup, ctes)

    if isinstance(cj, P.InSubquery):
        sub_df = _plan_select(cj.query, lookup, outer=None, ctes=ctes)
        sub_cols = sub_df.column_names()
        if len(sub_cols) != 1:
            raise SQLPlanError("IN subquery must produce one column")
        left = expr_to_daft(cj.child, binder)
        how = "anti" if cj.negated else "semi"
        return df.join(sub_df.distinct(), left_on=[left],
                       right_on=[col(sub_cols[0])], how=how)

    # comparison against a scalar subquery
    if isinstance(cj, P.BinOp) and cj.op in ("eq", "ne", "lt", "le", "gt",
                                             "ge"):
        sub_side = None
        if isinstance(cj.right, P.SubqueryExpr):
            sub_side, other = cj.right, cj.left
            op = cj.op
        elif isinstance(cj.left, P.SubqueryExpr):
            sub_side, other = cj.left, cj.right
            op = {"lt": "gt", "le": "ge", "gt": "lt", "ge": "le"}.get(
                cj.op, cj.op)
        if sub_side is not None:
            corr = _correlation_info(sub_side.query, binder, lookup, ctes)
            if corr is None:
                val = _eval_scalar_subquery(sub_side.query, binder, lookup,
                                            ctes)
                lhs = expr_to_daft(other, binder)
                cmp = getattr(lhs, {"eq": "__eq__", "ne": "__ne__",
                                    "lt": "__lt__", "le": "__le__",
                                    "gt": "__gt__", "ge": "__ge__"}[op])
                return df.where(cmp(lit(val)))
            return _plan_correlated_scalar(df, sub_side.query, corr, other,
                                           op, binder, lookup, ctes)
    raise SQLPlanError(f"unsupported subquery predicate: {cj!r}")


def _correlation_info(sub: P.SelectStmt, outer_binder: Binder, lookup,
                      ctes, allow_neq: bool = False):
    """Find equality conjuncts in sub.where referencing exactly one outer
    column and one inner column.  Returns (outer_cols_ast, inner_cols_ast,
    residual_where) or None if uncorrelated.  Raises on other correlation
    shapes.  With allow_neq, `inner <> outer` conjuncts are collected too
    and a 4-tuple (..., neq_pairs) is returned (the Q21 shape;
    decorrelated via per-group min/max in _plan_exists)."""
    if sub.where is None:
        return None
    inner_tables = {t.alias or t.name for t in sub.from_tables}
    inner_cols_set = set()
    refs_list = list(sub.from_tables) + [j.table for j in sub.joins]
    for t in refs_list:
        inner_tables.add(t.alias or t.name)
        try:
            if t.subquery is not None:
                continue
            src = ctes[t.name] if (ctes and t.name in ctes) \
                else lookup(t.name)
            inner_cols_set.update(src.column_names())
        except Exception:
            pass

    def is_outer_ref(r: P.Col) -> bool:
        if r.table is not None:
            if r.table in inner_tables:
                return False
            b = outer_binder
            while b is not None:
                if b.resolve(r.table, r.name) is not None:
                    return True
                b = b.outer
            return False
        if r.name in inner_cols_set:
            return False
        b = outer_binder
        while b is not None:
            try:
                if b.resolve(None, r.name) is not None:
                    return True
            except SQLPlanError:
                return True
            b = b.outer
        return False

    conjs = _split_conj(sub.where)
    outer_cols, inner_cols, residual, neq_pairs = [], [], [], []
    ineq_triples = []
    correlated = False
    for cj in conjs:
        refs: List[P.Col] = []
        _col_refs(cj, refs)
        outers = [r for r in refs if is_outer_ref(r)]
        if not outers:
            residual.append(cj)
            continue
        correlated = True
        _INEQ = ("lt", "le", "gt", "ge")
        ok = (isinstance(cj, P.BinOp) and
              (cj.op == "eq" or (allow_neq and cj.op == "ne") or
               (allow_neq and cj.op in _INEQ)) and
              isinstance(cj.left, P.Col) and isinstance(cj.right, P.Col))
        if not ok:
            raise SQLPlanError(
                f"unsupported correlated predicate: {cj!r} (only equality "
                "correlation is decorrelated)")
        if is_outer_ref(cj.left):
            pair = (cj.left, cj.right)
            op_outer_left = cj.op
        else:
            pair = (cj.right, cj.left)
            op_outer_left = {"lt": "gt", "le": "ge", "gt": "lt",
                             "ge": "le"}.get(cj.op, cj.op)
        if cj.op == "eq":
            outer_cols.append(pair[0])
            inner_cols.append(pair[1])
        elif cj.op in _INEQ:
            ineq_triples.append((pair[0], pair[1], op_outer_left))
        else:
            neq_pairs.append(pair)
    if not correlated:
        return None
    if allow_neq:
        return outer_cols, inner_cols, residual, neq_pairs, ineq_triples
    return outer_cols, inner_cols, residual


def _rebuild_where(conjs: List[Any]):
    if not conjs:
        return None
    out = conjs[0]
    for c in conjs[1:]:
        out = P.BinOp("and", out, c)
    return out


def _exists_parts(cj):
    """(sub_stmt, negated) if cj is EXISTS / NOT EXISTS, else None."""
    if isinstance(cj, P.UnaryOp) and cj.op == "not" and \
            isinstance(cj.child, P.ExistsExpr):
        return cj.child.query, not cj.child.negated
    if isinstance(cj, P.ExistsExpr):
        return cj.query, cj.negated
    return None


def _ast_map_cols(node, fn):
    """Recursively rewrite P.Col nodes of a dataclass AST."""
    import dataclasses as _dc
    if isinstance(node, P.Col):
        out = fn(node)
        return out if out is not None else node
    if _dc.is_dataclass(node) and not isinstance(node, type):
        changes = {}
        for f in _dc.fields(node):
            v = getattr(node, f.name)
            nv = _ast_map_val(v, fn)
            if nv is not v:
                changes[f.name] = nv
        return _dc.replace(node, **changes) if changes else node
    return node


def _ast_map_val(v, fn):
    import dataclasses as _dc
    if isinstance(v, list):
        nl = [_ast_map_val(x, fn) for x in v]
        return nl if any(a is not b for a, b in zip(nl, v)) else v
    if isinstance(v, tuple):
        nt = tuple(_ast_map_val(x, fn) for x in v)
        return nt if any(a is not b for a, b in zip(nt, v)) else v
    if _dc.is_dataclass(v) and not isinstance(v, type):
        return _ast_map_cols(v, fn)
    return v


def _sub_alias_map(sub: P.SelectStmt) -> dict:
    m = {}
    for t in list(sub.from_tables) + [j.table for j in sub.joins]:
        m[t.alias or t.name] = t.name
    return m


def _canon(ast, alias2tbl: dict) -> str:
    """Alias-insensitive repr: table qualifiers replaced by table names."""
    return repr(_ast_map_cols(
        ast, lambda c: P.Col(alias2tbl.get(c.table, c.table), c.name)))


def _try_fuse_exists_pair(df, subq_conjs, binder, lookup, ctes):
    """Fuse two EXISTS-with-<> conjuncts over the same correlated group
    (same FROM, same correlation keys, same <> column) whose residual
    predicates are ordered by implication (R1 ⊆ R2) into ONE
    groupby(min/max + filter-masked min/max) and ONE left join — the
    single-pass q21 plan the round-1 benchmark hand-wrote
    (benchmarks/tpch/queries.py q21), now derived by the planner."""
    cands = []
    for cj in subq_conjs:
        parts = _exists_parts(cj)
        if parts is None:
            continue
        sub, negated = parts
        try:
            corr = _correlation_info(sub, binder, lookup, ctes,
                                     allow_neq=True)
        except SQLPlanError:
            continue
        if corr is None or len(corr) != 5 or not corr[3]:
            continue
        outer_cols, inner_cols, residual, neq_pairs, ineq_triples = corr
        if len(neq_pairs) != 1 or not inner_cols or ineq_triples:
            continue
        cands.append((cj, sub, negated, outer_cols, inner_cols,
                      residual, neq_pairs[0]))
    for i in range(len(cands)):
        for j in range(len(cands)):
            if i == j:
                continue
            a, b = cands[i], cands[j]
            am, bm = _sub_alias_map(a[1]), _sub_alias_map(b[1])
            # same tables (alias-insensitively), same joins
            if [t.name for t in a[1].from_tables] != \
                    [t.name for t in b[1].from_tables]:
                continue
            if len(a[1].joins) != len(b[1].joins) or any(
                    ja.table.name != jb.table.name or ja.how != jb.how or
                    _canon(ja.on, am) != _canon(jb.on, bm)
                    for ja, jb in zip(a[1].joins, b[1].joins)):
                continue
            # same correlation keys and same <> pair
            if [_canon(c, am) for c in a[4]] != \
                    [_canon(c, bm) for c in b[4]]:
                continue
            if _canon(a[6][0], am) != _canon(b[6][0], bm) or \
                    _canon(a[6][1], am) != _canon(b[6][1], bm):
                continue
            ra = {_canon(c, am) for c in a[5]}
            rbl = [(_canon(c, bm), c) for c in b[5]]
            if not ra <= {k for k, _ in rbl}:
                continue
            # re-qualify the extra conjuncts into a's aliases
            b2a = {(tb.alias or tb.name): (ta.alias or ta.name)
                   for ta, tb in zip(a[1].from_tables, b[1].from_tables)}
            for ja, jb in zip(a[1].joins, b[1].joins):
                b2a[jb.table.alias or jb.table.name] = \
                    ja.table.alias or ja.table.name
            extra = [_ast_map_cols(
                c, lambda cc: P.Col(b2a.get(cc.table, cc.table), cc.name))
                for k, c in rbl if k not in ra]
            return _plan_fused_exists(df, a, b, extra, binder, lookup,
                                      ctes), [a[0], b[0]]
    return None


def _plan_fused_exists(df, base, filt, extra, binder, lookup, ctes):
    """base = the EXISTS with the weaker residual R1; filt = the one with
    R1 ∪ extra.  One subquery over R1 computes min/max of the <> column
    per key plus mask-filtered min/max (mask = AND(extra))."""
    _cj1, sub, neg1, outer_cols, inner_cols, residual, (outer_b, inner_b) \
        = base
    neg2 = filt[2]
    items = [P.SelectItem(c, f"__ex{i}") for i, c in enumerate(inner_cols)]
    items.append(P.SelectItem(inner_b, "__exb"))
    if extra:
        mask = extra[0]
        for e in extra[1:]:
            mask = P.BinOp("and", mask, e)
        items.append(P.SelectItem(mask, "__exm"))
    inner_stmt = P.SelectStmt(
        items=items, from_tables=sub.from_tables, joins=sub.joins,
        where=_rebuild_where(residual))
    sub_df = _plan_select(inner_stmt, lookup, outer=None, ctes=ctes)
    if extra:
        sub_df = sub_df.with_column(
            "__exbm", col("__exm").if_else(col("__exb"), lit(None)))
    else:
        sub_df = sub_df.with_column("__exbm", col("__exb"))
    keys = [f"__ex{i}" for i in range(len(inner_cols))]
    agg_df = sub_df.groupby(*keys).agg(
        col("__exb").min().alias("__exmn"),
        col("__exb").max().alias("__exmx"),
        col("__exbm").min().alias("__exln"),
        col("__exbm").max().alias("__exlx"))
    orig = df.column_names()
    left = [expr_to_daft(c, binder) for c in outer_cols]
    joined = df.join(agg_df, left_on=left,
                     right_on=[col(k) for k in keys], how="left")
    ob = expr_to_daft(outer_b, binder)

    def pred_for(mn, mx, negated):
        if negated:
            return mn.is_null() | ob.is_null() | ((mn == ob) & (mx == ob))
        return mn.is_null().if_else(lit(False), (mn != ob) | (mx != ob))

    pred = pred_for(col("__exmn"), col("__exmx"), neg1) & \
        pred_for(col("__exln"), col("__exlx"), neg2)
    return joined.where(pred).select(*orig)


def _plan_exists(df, sub: P.SelectStmt, negated: bool, binder, lookup, ctes):
    corr = _correlation_info(sub, binder, lookup, ctes, allow_neq=True)
    if corr is None:
        # uncorrelated EXISTS: keep/drop everything
        sub_df = _plan_select(sub, lookup, outer=None, ctes=ctes)
        n = sub_df.limit(1).count_rows()
        keep = (n > 0) != negated
        return df if keep else df.where(lit(False))
    outer_cols, inner_cols, residual, neq_pairs, ineq_triples = corr
    if ineq_triples:
        if neq_pairs:
            raise SQLPlanError("EXISTS with mixed <> and range "
                               "correlation is not supported")
        return _plan_exists_ineq(df, sub, negated, binder, lookup, ctes,
                                 outer_cols, inner_cols, residual,
                                 ineq_triples)
    if neq_pairs:
        return _plan_exists_neq(df, sub, negated, binder, lookup, ctes,
                                outer_cols, inner_cols, residual, neq_pairs)
    inner_stmt = P.SelectStmt(
        items=[P.SelectItem(c, f"__ex{i}") for i, c in enumerate(inner_cols)],
        from_tables=sub.from_tables, joins=sub.joins,
        where=_rebuild_where(residual))
    sub_df = _plan_select(inner_stmt, lookup, outer=None, ctes=ctes)
    left = [expr_to_daft(c, binder) for c in outer_cols]
    right = [col(f"__ex{i}") for i in range(len(inner_cols))]
    how = "anti" if negated else "semi"
    return df.join(sub_df, left_on=left, right_on=right, how=how)


def _plan_exists_ineq(df, sub, negated, binder, lookup, ctes,
                      outer_cols, inner_cols, residual, ineq_triples):
    """EXISTS with range correlation beside (optional) equality keys:

        EXISTS(SELECT 1 FROM u WHERE u.k = t.k AND u.w < t.v AND resid)

    decorrelates via a row-id semi join: tag outer rows, inner-join on
    the equality keys, filter the range predicates on the joined frame,
    and semi/anti join the distinct matching row ids back (duckdb-style
    general unnesting; ref: daft subquery unnesting rules)."""
    rid = "__xrid"
    dfr = df.add_monotonically_increasing_id(rid)
    keys = [f"__exk{i}" for i in range(len(inner_cols))]
    ivs = [f"__exv{i}" for i in range(len(ineq_triples))]
    inner_stmt = P.SelectStmt(
        items=[P.SelectItem(c, k) for c, k in zip(inner_cols, keys)] +
              [P.SelectItem(t[1], v)
               for t, v in zip(ineq_triples, ivs)],
        from_tables=sub.from_tables, joins=sub.joins,
        where=_rebuild_where(residual))
    sub_df = _plan_select(inner_stmt, lookup, outer=None, ctes=ctes)
    if keys:
        joined = dfr.join(sub_df,
                          left_on=[expr_to_daft(c, binder)
                                   for c in outer_cols],
                          right_on=[col(k) for k in keys], how="inner")
    else:
        # no equality keys: constant-key join (cross product)
        joined = dfr.with_column("__one", lit(1)).join(
            sub_df.with_column("__one2", lit(1)),
            left_on=[col("__one")], right_on=[col("__one2")],
            how="inner")
    for (outer_ast, _inner_ast, op), v in zip(ineq_triples, ivs):
        lhs = expr_to_daft(outer_ast, binder)
        cmp = getattr(lhs, {"lt": "__lt__", "le": "__le__",
                            "gt": "__gt__", "ge": "__ge__"}[op])
        joined = joined.where(cmp(col(v)))
    matched = joined.select(col(rid)).distinct()
    how = "anti" if negated else "semi"
    out = dfr.join(matched, left_on=[col(rid)], right_on=[col(rid)],
                   how=how)
    return out.select(*[col(c) for c in df.column_names()])


def _plan_exists_neq(df, sub, negated, binder, lookup, ctes,
                     outer_cols, inner_cols, residual, neq_pairs):
    """EXISTS with one `inner.b <> outer.b` correlated conjunct beside the
    equality correlation (TPC-H Q21):

        EXISTS(SELECT .. WHERE in.a = out.a AND in.b <> out.b AND resid)
          <=>  the b-set of group a (after resid) is non-empty and not
               exactly {out.b}  <=>  min(b) != out.b OR max(b) != out.b

    so the subquery collapses to GROUP BY a AGG min(b), max(b) + one left
    join (no per-row re-execution).  Ref behavior:
    /root/reference/src/daft-sql (EXISTS planning) + duckdb-style
    decorrelation of anti-dependent predicates."""
    if len(neq_pairs) != 1:
        raise SQLPlanError("EXISTS with more than one <> correlated "
                           "conjunct is not supported")
    if not inner_cols:
        raise SQLPlanError("EXISTS with <> correlation requires at least "
                           "one equality correlation conjunct")
    outer_b_ast, inner_b_ast = neq_pairs[0]
    items = [P.SelectItem(c, f"__ex{i}") for i, c in enumerate(inner_cols)]
    items.append(P.SelectItem(inner_b_ast, "__exb"))
    inner_stmt = P.SelectStmt(
        items=items, from_tables=sub.from_tables, joins=sub.joins,
        where=_rebuild_where(residual))
    sub_df = _plan_select(inner_stmt, lookup, outer=None, ctes=ctes)
    keys = [f"__ex{i}" for i in range(len(inner_cols))]
    agg_df = sub_df.groupby(*keys).agg(
        col("__exb").min().alias("__exmn"),
        col("__exb").max().alias("__exmx"))
    orig = df.column_names()
    left = [expr_to_daft(c, binder) for c in outer_cols]
    joined = df.join(agg_df, left_on=left,
                     right_on=[col(k) for k in keys], how="left")
    ob = expr_to_daft(outer_b_ast, binder)
    mn, mx = col("__exmn"), col("__exmx")
    if negated:
        # NOT EXISTS: group empty, or its b-set is exactly {out.b}
        pred = mn.is_null() | ob.is_null() | \
            ((mn == ob) & (mx == ob))
    else:
        pred = mn.is_null().if_else(lit(False),
                                    (mn != ob) | (mx != ob))
    return joined.where(pred).select(*orig)


def _plan_correlated_scalar(df, sub: P.SelectStmt, corr, other_ast, op,
                            binder, lookup, ctes):
    """Decorrelate `expr <op> (SELECT agg(...) FROM .. WHERE inner = outer)`
    into groupby + join (the Q2/Q17 rewrite)."""
    outer_cols, inner_cols, residual = corr
    if len(sub.items) != 1 or sub.items[0].star:
        raise SQLPlanError("scalar subquery must select one aggregate")
    agg_ast = sub.items[0].expr
    if not _has_aggregate(agg_ast):
        raise SQLPlanError("correlated scalar subquery must aggregate")
    inner_stmt = P.SelectStmt(
        items=[P.SelectItem(c, f"__sk{i}")
               for i, c in enumerate(inner_cols)] +
              [P.SelectItem(agg_ast, "__sv")],
        from_tables=sub.from_tables, joins=sub.joins,
        where=_rebuild_where(residual),
        group_by=list(inner_cols))
    sub_df = _plan_select(inner_stmt, lookup, outer=None, ctes=ctes)
    left = [expr_to_daft(c, binder) for c in outer_cols]
    right = [col(f"__sk{i}") for i in range(len(inner_cols))]
    joined = df.join(sub_df, left_on=left, right_on=right, how="inner")
    lhs = expr_to_daft(other_ast, binder)
    cmp = getattr(lhs, {"eq": "__eq__", "ne": "__ne__", "lt": "__lt__",
                        "le": "__le__", "gt": "__gt__", "ge": "__ge__"}[op])
    out = joined.where(cmp(col("__sv")))
    keep = [c for c in out.column_names()
            if not c.startswith("__sk") and c != "__sv"]
    return out.select(*keep)
