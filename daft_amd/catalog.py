"""Catalog / Session (ref: /root/reference/src/daft-catalog/ in-memory impl +
daft-session Session with attached catalogs, temp tables, options)."""
from __future__ import annotations

import threading
from typing import Dict, List, Optional


# catalog property bags and errors (ref: daft/catalog/__init__.py:73-76)
Properties = dict


class NotFoundError(Exception):
    """Raised when a catalog object (table/namespace/function) is
    missing (ref: daft.catalog.NotFoundError)."""


class Function:
    """A catalog-registered function: calling it with expression
    arguments produces an Expression (ref: daft/catalog Function ABC)."""

    def __init__(self, identifier, fn=None):
        self._identifier = identifier if isinstance(identifier, Identifier) \
            else Identifier(str(identifier))
        self._fn = fn

    @property
    def identifier(self):
        return self._identifier

    @property
    def name(self):
        return self._identifier.parts[-1]

    @property
    def namespace(self):
        return Identifier(*self._identifier.parts[:-1]) \
            if len(self._identifier.parts) > 1 else Identifier()

    def __call__(self, *args, **kwargs):
        if self._fn is None:
            raise NotFoundError(f"function {self.name} has no binding")
        return self._fn(*args, **kwargs)


class Identifier:
    """Dotted table identifier (ref: daft-catalog Identifier)."""

    def __init__(self, *parts: str):
        self.parts = list(parts)

    @staticmethod
    def parse(text: str) -> "Identifier":
        return Identifier(*text.split("."))

    def __str__(self):
        return ".".join(self.parts)

    def __eq__(self, o):
        return isinstance(o, Identifier) and o.parts == self.parts

    def __hash__(self):
        return hash(tuple(self.parts))


class Catalog:
    """Catalog ABC (ref: daft-catalog Catalog trait)."""

    name: str = "catalog"

    def list_tables(self, pattern: Optional[str] = None) -> List[str]:
        raise NotImplementedError

    def get_table(self, name: str):
        raise NotImplementedError

    def create_table(self, name: str, df) -> None:
        raise NotImplementedError

    def drop_table(self, name: str) -> None:
        raise NotImplementedError


class MemoryCatalog(Catalog):
    """In-memory catalog (ref: daft-catalog/src/impls/memory.rs)."""

    def __init__(self, name: str = "memory"):
        self.name = name
        self._tables: Dict[str, object] = {}
        self._lock = threading.Lock()

    def list_tables(self, pattern: Optional[str] = None) -> List[str]:
        with self._lock:
            names = sorted(self._tables)
        if pattern:
            names = [n for n in names if pattern in n]
        return names

    def get_table(self, name: str):
        with self._lock:
            if name not in self._tables:
                raise KeyError(f"table {name!r} not found in {self.name}")
            return self._tables[name]

    def create_table(self, name: str, df) -> None:
        with self._lock:
            self._tables[name] = df

    def drop_table(self, name: str) -> None:
        with self._lock:
            self._tables.pop(name, None)


class Session:
    """Query session: attached catalogs + temp tables + options (ref:
    daft-session/src/session.rs)."""

    def __init__(self):
        self._catalogs: Dict[str, Catalog] = {}
        self._current: Optional[str] = None
        self._temp = MemoryCatalog("temp")
        self.options: Dict[str, object] = {}

    def attach_catalog(self, catalog: Catalog,
                       alias: Optional[str] = None) -> None:
        name = alias or catalog.name
        self._catalogs[name] = catalog
        if self._current is None:
            self._current = name

    def detach_catalog(self, name: str) -> None:
        self._catalogs.pop(name, None)
        if self._current == name:
            self._current = next(iter(self._catalogs), None)

    def set_catalog(self, name: str) -> None:
        if name not in self._catalogs:
            raise KeyError(name)
        self._current = name

    def current_catalog(self) -> Optional[Catalog]:
        return self._catalogs.get(self._current) if self._current else None

    def create_temp_table(self, name: str, df) -> None:
        self._temp.create_table(name, df)

    def list_tables(self) -> List[str]:
        out = list(self._temp.list_tables())
        for c in self._catalogs.values():
            out.extend(f"{c.name}.{t}" for t in c.list_tables())
        return out

    def get_table(self, name):
        ident = Identifier.parse(name) if isinstance(name, str) else name
        if len(ident.parts) == 1:
            try:
                return self._temp.get_table(ident.parts[0])
            except KeyError:
                cat = self.current_catalog()
                if cat is not None:
                    return cat.get_table(ident.parts[0])
                raise
        cat = self._catalogs[ident.parts[0]]
        return cat.get_table(".".join(ident.parts[1:]))

    def sql(self, query: str):
        """Run SQL against the session's tables.  Beside SELECT, a small
        DDL surface is supported (capability of the reference's
        Session.sql): CREATE [OR REPLACE] [TEMP] TABLE name AS SELECT,
        DROP TABLE name, SHOW TABLES."""
        import re
        from .sql.planner import plan_sql
        q = query.strip().rstrip(";")
        m = re.match(
            r"(?is)^create\s+(or\s+replace\s+)?(temp(?:orary)?\s+)?table"
            r"\s+([\w.\"]+)\s+as\s+(.*)$", q)
        if m:
            replace = m.group(1) is not None
            name = m.group(3).strip('"')
            df = plan_sql(m.group(4), lambda n: self.get_table(n))
            if not replace and name in self._temp.list_tables():
                raise ValueError(f"table {name!r} already exists "
                                 f"(use CREATE OR REPLACE)")
            self.create_temp_table(name, df.collect())
            return df
        m = re.match(r"(?is)^drop\s+table\s+(if\s+exists\s+)?([\w.\"]+)$",
                     q)
        if m:
            name = m.group(2).strip('"')
            if not m.group(1) and name not in self._temp.list_tables():
                raise KeyError(name)
            self._temp.drop_table(name)
            return None
        if re.match(r"(?is)^show\s+tables$", q):
            from . import from_pydict
            return from_pydict({"table": self.list_tables()})
        return plan_sql(q, lambda n: self.get_table(n))


_session: Optional[Session] = None
_session_lock = threading.Lock()


def current_session() -> Session:
    global _session
    with _session_lock:
        if _session is None:
            _session = Session()
        return _session
