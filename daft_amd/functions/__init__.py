"""Function library (ref: /root/reference/daft/functions/ and
src/daft-functions*)."""
from __future__ import annotations

from typing import Optional

from ..expressions.expressions import (Coalesce, Expression, _to_node,
                                       ScalarFn)
from ..schema import DataType


def coalesce(*exprs) -> Expression:
    return Expression(Coalesce([_to_node(e) for e in exprs]))


def row_number() -> Expression:
    from ..physical.window import WindowFn
    return Expression(WindowFn("row_number", None, None))


def rank() -> Expression:
    from ..physical.window import WindowFn
    return Expression(WindowFn("rank", None, None))


def dense_rank() -> Expression:
    from ..physical.window import WindowFn
    return Expression(WindowFn("dense_rank", None, None))


def monotonically_increasing_id() -> Expression:
    raise NotImplementedError(
        "use DataFrame.add_monotonically_increasing_id()")


def columns_sum(*exprs) -> Expression:
    out = _expr(exprs[0])
    for e in exprs[1:]:
        out = out + _expr(e)
    return out


def columns_avg(*exprs) -> Expression:
    return columns_sum(*exprs) / float(len(exprs))


def columns_min(*exprs) -> Expression:
    out = _expr(exprs[0])
    for e in exprs[1:]:
        nxt = _expr(e)
        out = (out <= nxt).if_else(out, nxt)
    return out


def columns_max(*exprs) -> Expression:
    out = _expr(exprs[0])
    for e in exprs[1:]:
        nxt = _expr(e)
        out = (out >= nxt).if_else(out, nxt)
    return out


def _expr(e) -> Expression:
    from ..expressions.expressions import col
    return col(e) if isinstance(e, str) else e


def cosine_distance(a, b) -> Expression:
    return _expr(a).embedding.cosine_distance(_expr(b))


def uuid() -> Expression:
    import uuid as _uuid
    from ..series import Series

    def gen(s) -> Series:
        vals = [str(_uuid.uuid4()) for _ in range(len(s))]
        out = Series.from_pylist("uuid", vals, DataType.string())
        return out.to(s.device) if s.is_gpu() else out
    from ..expressions.expressions import ColumnRef
    raise NotImplementedError("uuid() requires a column context; "
                              "use df.add_monotonically_increasing_id")

from .spatial import great_circle_distance  # noqa: E402,F401


def file(expr) -> "Expression":
    """Wrap a path (string) or bytes column as File objects (ref
    capability: daft.functions.file / daft.File columns)."""
    from ..expressions.expressions import Expression, ScalarFn, _to_node
    from ..schema import DataType
    from ..series import Series
    from ..file import File

    def run(s):
        vals = s.cpu().to_pylist()
        objs = [None if v is None else File(v) for v in vals]
        return Series(s.name, DataType.python(), pyobjs=objs,
                      validity=None, length=len(vals))
    return Expression(ScalarFn("file", run, [_to_node(expr)],
                               DataType.python()))


def file_size(expr) -> "Expression":
    """Size in bytes of a File column (ref: daft/functions/file_.py:110)."""
    from ..expressions.expressions import Expression, ScalarFn, _to_node
    from ..schema import DataType
    from ..series import Series

    def run(s):
        out = [None if f is None else f.size() for f in s.pyobjs]
        return Series.from_pylist(s.name, out, DataType.int64())
    return Expression(ScalarFn("file_size", run, [_to_node(expr)],
                               DataType.int64()))
