from .expressions import (  # noqa: F401
    Expression, ExprNode, col, lit, element, interval, list_, struct,
    AggKind, resolve_exprs,
)
