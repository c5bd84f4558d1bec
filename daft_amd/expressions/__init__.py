from .expressions import (  # noqa: F401
    Expression, ExprNode, col, lit, element, interval, list_, struct,
    AggKind, resolve_exprs,
)
from .visitor import ExpressionVisitor, ExpressionsProjection  # noqa: E402,F401
from ..functions.misc import _WhenThen as WhenExpr  # noqa: E402,F401
