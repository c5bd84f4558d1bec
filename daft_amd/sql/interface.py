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
    q = query.strip()
    if q.lower().startswith("explain "):
        df = plan_sql(q[8:], lambda n: _lookup(n, catalog, frame_vars))
        text = df._builder.optimize().explain()
        from ..io import from_pydict
        return from_pydict({"plan": text.splitlines()})
    return plan_sql(q, lambda n: _lookup(n, catalog, frame_vars))


def sql_expr(text: str):
    """Parse a scalar SQL expression into an Expression."""
    from .parser import parse_expression
    return parse_expression(text)
