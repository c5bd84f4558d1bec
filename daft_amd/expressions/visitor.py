"""ExpressionVisitor — visitor pattern over expression trees (ref:
/root/reference/daft/expressions/visitor.py).  Subclass and implement
visit_col / visit_lit / visit_alias / visit_cast / visit_function;
`visit()` dispatches on the node kind, `generic_visit` recurses."""
from __future__ import annotations

from abc import ABC
from typing import Any, Generic, TypeVar

from .expressions import (Agg, Alias, BinaryOp, Cast, ColumnRef, Expression,
                          ExprNode, Literal)

R = TypeVar("R")


class ExpressionVisitor(ABC, Generic[R]):
    def visit(self, expr) -> R:
        node = expr._node if isinstance(expr, Expression) else expr
        if isinstance(node, ColumnRef):
            return self.visit_col(node.name)
        if isinstance(node, Literal):
            return self.visit_lit(node.value)
        if isinstance(node, Alias):
            return self.visit_alias(Expression(node.child), node.name)
        if isinstance(node, Cast):
            return self.visit_cast(Expression(node.child), node.dtype)
        name = type(node).__name__.lower()
        if isinstance(node, BinaryOp):
            name = node.op
        elif isinstance(node, Agg):
            name = node.kind.value
        args = [Expression(c) for c in node.children()]
        return self.visit_function(name, args)

    def generic_visit(self, expr) -> None:
        node = expr._node if isinstance(expr, Expression) else expr
        for c in node.children():
            self.visit(Expression(c))

    # default hooks delegate to visit_function / generic recursion;
    # subclasses override what they need
    def visit_col(self, name: str) -> R:
        return self.visit_function("col", [])

    def visit_lit(self, value: Any) -> R:
        return self.visit_function("lit", [])

    def visit_alias(self, expr, alias: str) -> R:
        return self.visit(expr)

    def visit_cast(self, expr, dtype) -> R:
        return self.visit(expr)

    def visit_function(self, name: str, args) -> R:
        for a in args:
            self.visit(a)
        return None  # type: ignore[return-value]


class ExpressionsProjection:
    """An ordered collection of uniquely-named Expressions (ref:
    daft ExpressionsProjection)."""

    def __init__(self, exprs):
        self._exprs = list(exprs)
        names = [e._node.out_name() for e in self._exprs]
        if len(set(names)) != len(names):
            raise ValueError("duplicate names in ExpressionsProjection")

    @classmethod
    def from_schema(cls, schema) -> "ExpressionsProjection":
        from .expressions import col
        return cls([col(f.name) for f in schema])

    def __iter__(self):
        return iter(self._exprs)

    def __len__(self):
        return len(self._exprs)

    def to_name_set(self):
        return {e._node.out_name() for e in self._exprs}

    def union(self, other, rename_dup=None) -> "ExpressionsProjection":
        from .expressions import Expression
        out = list(self._exprs)
        seen = self.to_name_set()
        for e in other:
            n = e._node.out_name()
            if n in seen:
                if rename_dup is None:
                    raise ValueError(f"duplicate name {n}")
                e = e.alias(rename_dup + n)
            out.append(e)
            seen.add(e._node.out_name())
        return ExpressionsProjection(out)
