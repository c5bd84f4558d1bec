"""SQL entry points (ref: /root/reference/src/daft-sql/src/planner.rs and
daft/sql/).  The parser/planner lives in parser.py / planner.py."""
from __future__ import annotations

import threading
from typing import Dict, Optional

_registry_lock = threading.Lock()
_tables: Dict[str, "object"] = {}


class SQLCatalog:
    """Simple table catalog for daft_amd.sql (ref: daft-session catalogs)."""

    def __init__(self, tables: Optional[dict] = None):
        self.tables = dict(tables or {})

    def register_table(self, name: str, df) -> None:
        self.tables[name] = df

    def get_table(self, name: str):
        if name in self.tables:
            return self.tables[name]
        raise KeyError(f"table {name} not found in catalog")


def register_table(name: str, df) -> None:
    with _registry_lock:
        _tables[name] = df


def _lookup(name: str, catalog: Optional[SQLCatalog], frame_vars: dict):
    if catalog is not None:
        try:
            return catalog.get_table(name)
        except KeyError:
            pass
    with _registry_lock:
        if name in _tables:
            return _tables[name]
    v = frame_vars.get(name)
    from ..dataframe import DataFrame
    if isinstance(v, DataFrame):
        return v
    # ambient session tables (CREATE TABLE ... AS, create_temp_table)
    try:
        from ..session_api import current_session
        t = current_session().get_table(name)
        if t is not None:
            return t
    except Exception:
        pass
    raise KeyError(f"unknown table {name!r} in SQL query")


def sql(query: str, catalog: Optional[SQLCatalog] = None, **kwargs):
    """Run a SQL query against registered tables / DataFrame variables in the
    caller's scope (matches reference `daft.sql` ergonomics)."""
    import inspect
    frame = inspect.currentframe().f_back
    frame_vars = {}
    if frame is not None:
        frame_vars.update(frame.f_globals)
        frame_vars.update(frame.f_locals)
    from .planner import plan_sql
    q = query.strip().rstrip(";")
    lookup = lambda n: _lookup(n, catalog, frame_vars)  # noqa: E731
    stmt = _try_statement(q, lookup)
    if stmt is not None:
        return stmt
    if q.lower().startswith("explain "):
        df = plan_sql(q[8:], lookup)
        text = df._builder.optimize().explain()
        from ..io import from_pydict
        return from_pydict({"plan": text.splitlines()})
    return plan_sql(q, lookup)


def _try_statement(q: str, lookup):
    """Non-SELECT statements the reference's SQL layer supports
    (ref: daft-sql/src/statement.rs — ShowTables, Use, CreateTable,
    Describe; plus DROP TABLE for symmetry)."""
    import re
    from .planner import plan_sql
    from ..io import from_pydict
    low = q.lower()
    if low.startswith("show tables"):
        from ..session_api import list_tables
        m = re.match(r"show tables(?:\s+like\s+'([^']*)')?\s*$", low)
        pat = m.group(1) if m else None
        names = list_tables()
        if pat:
            rx = re.compile("^" + pat.replace("%", ".*")
                            .replace("_", ".") + "$")
            names = [n for n in names if rx.match(n)]
        return from_pydict({"table": list(names)} if names
                           else {"table": []})
    m = re.match(r"use\s+([A-Za-z_][\w.]*)\s*$", low)
    if m:
        from ..session_api import current_session
        current_session().options["current_namespace"] = m.group(1)
        return from_pydict({"ok": [True]})
    m = re.match(r"create\s+(?:or\s+replace\s+)?(?:temp(?:orary)?\s+)?"
                 r"table\s+([A-Za-z_]\w*)\s+as\s+(.*)$", q,
                 re.IGNORECASE | re.DOTALL)
    if m:
        name, body = m.group(1), m.group(2)
        df = plan_sql(body, lookup).collect()
        from ..session_api import create_temp_table
        create_temp_table(name, df)
        return df
    m = re.match(r"drop\s+table\s+(?:if\s+exists\s+)?([A-Za-z_]\w*)\s*$",
                 q, re.IGNORECASE)
    if m:
        from ..session_api import current_session, drop_table
        try:
            drop_table(m.group(1))
        except Exception:
            if "if exists" not in q.lower():
                raise
        return from_pydict({"ok": [True]})
    m = re.match(r"(?:describe|desc)\s+(.+)$", q,
                 re.IGNORECASE | re.DOTALL)
    if m:
        target = m.group(1).strip()
        if re.match(r"^[A-Za-z_]\w*$", target):
            df = lookup(target)
        else:
            df = plan_sql(target, lookup)
        sch = df.schema
        return from_pydict({"column_name": [f.name for f in sch],
                            "type": [str(f.dtype) for f in sch]})
    return None


def sql_expr(text: str):
    """Parse a scalar SQL expression into an Expression."""
    from .parser import parse_expression
    return parse_expression(text)
