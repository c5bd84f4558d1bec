"""Expression IR + vectorized evaluation.

The analog of the reference's daft-dsl Expr enum
(/root/reference/src/daft-dsl/src/expr/mod.rs:222-330): Column, Alias,
Literal, BinaryOp, Cast, Not/IsNull/NotNull/FillNull, IsIn, Between, IfElse,
Agg, ScalarFn, Coalesce, plus python UDFs.  Evaluation maps an expression
tree over a RecordBatch into a Series, launching GPU kernels column-at-a-time
(ref eval: daft-recordbatch/src/lib.rs:875 eval_expression_list).
"""
from __future__ import annotations

import datetime as _dt
from typing import Any, Callable, Dict, List, Optional, Sequence, Tuple, Union

import torch

from ..schema import DataType, Field, Schema, TypeKind, supertype
from ..series import Series, lit_series

# ---------------------------------------------------------------------------
# nodes
# ---------------------------------------------------------------------------


class ExprNode:
    """Base expression node; immutable."""

    def children(self) -> List["ExprNode"]:
        return []

    def with_children(self, ch: List["ExprNode"]) -> "ExprNode":
        assert not ch
        return self

    # output name resolution (daft semantics: leftmost column name)
    def out_name(self) -> str:
        ch = self.children()
        if ch:
            return ch[0].out_name()
        return "literal"

    def to_field(self, schema: Schema) -> Field:
        raise NotImplementedError(type(self))

    def evaluate(self, batch) -> Series:
        raise NotImplementedError(type(self))

    def is_aggregation(self) -> bool:
        """Does this subtree contain an Agg node?"""
        return any(c.is_aggregation() for c in self.children())

    def column_refs(self) -> List[str]:
        out: List[str] = []
        seen = set()

        def rec(n: ExprNode):
            if isinstance(n, ColumnRef):
                if n.name not in seen:
                    seen.add(n.name)
                    out.append(n.name)
            for c in n.children():
                rec(c)
        rec(self)
        return out

    def semantic_id(self) -> str:
        """Canonical string for plan fingerprinting / CSE."""
        return repr(self)


class ColumnRef(ExprNode):
    def __init__(self, name: str):
        self.name = name

    def out_name(self) -> str:
        return self.name

    def to_field(self, schema: Schema) -> Field:
        return schema[self.name]

    def evaluate(self, batch) -> Series:
        return batch.column(self.name)

    def __repr__(self):
        return f"col({self.name})"


class Literal(ExprNode):
    def __init__(self, value: Any, dtype: Optional[DataType] = None):
        self.value = value
        if dtype is None:
            from ..series import _infer_dtype
            dtype = _infer_dtype([value])
        self.dtype = dtype

    def to_field(self, schema: Schema) -> Field:
        return Field("literal", self.dtype)

    def evaluate(self, batch) -> Series:
        # stays length-1: elementwise consumers broadcast without
        # materializing a full column (operators that need full length
        # broadcast explicitly)
        return lit_series("literal", self.value, self.dtype,
                          device=batch.device)

    def __repr__(self):
        # dtype is part of node identity: CSE/agg-decomposition key subtrees
        # by repr, and lit(30):int32 must not merge with lit(30):int64
        return f"lit({self.value!r}:{self.dtype!r})"


class Alias(ExprNode):
    def __init__(self, child: ExprNode, name: str):
        self.child = child
        self.name = name

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return Alias(ch[0], self.name)

    def out_name(self) -> str:
        return self.name

    def to_field(self, schema: Schema) -> Field:
        return Field(self.name, self.child.to_field(schema).dtype)

    def evaluate(self, batch) -> Series:
        return self.child.evaluate(batch).rename(self.name)

    def is_aggregation(self):
        return self.child.is_aggregation()

    def __repr__(self):
        return f"{self.child!r}.alias({self.name})"


_ARITH = {"add", "sub", "mul", "div", "floordiv", "mod", "pow"}
_CMP = {"eq", "ne", "lt", "le", "gt", "ge"}
_LOGIC = {"and", "or", "xor"}


class BinaryOp(ExprNode):
    def __init__(self, op: str, left: ExprNode, right: ExprNode):
        self.op = op
        self.left = left
        self.right = right

    def children(self):
        return [self.left, self.right]

    def with_children(self, ch):
        return BinaryOp(self.op, ch[0], ch[1])

    def to_field(self, schema: Schema) -> Field:
        lf = self.left.to_field(schema)
        rf = self.right.to_field(schema)
        if self.op in _CMP or self.op in _LOGIC:
            return Field(lf.name, DataType.bool())
        if self.op == "div":
            return Field(lf.name, DataType.float64())
        if self.op == "sub" and lf.dtype.kind == rf.dtype.kind == TypeKind.DATE:
            return Field(lf.name, DataType.duration("us"))
        if lf.dtype.is_temporal():
            return Field(lf.name, lf.dtype)
        if lf.dtype.is_string() and self.op == "add":
            return Field(lf.name, DataType.string())
        if lf.dtype.is_decimal() or rf.dtype.is_decimal():
            from ..schema import decimal_binary_result

            def _digs(node):
                if isinstance(node, Literal) and \
                        isinstance(node.value, int) and \
                        not isinstance(node.value, bool):
                    return max(1, len(str(abs(node.value))))
                return None
            out = decimal_binary_result(lf.dtype, rf.dtype, self.op,
                                        _digs(self.left),
                                        _digs(self.right))
            if out is not None:
                return Field(lf.name, out)
        return Field(lf.name, supertype(lf.dtype, rf.dtype))

    def evaluate(self, batch) -> Series:
        l = self.left.evaluate(batch)
        r = self.right.evaluate(batch)
        if self.op in _CMP:
            return l.compare(r, self.op)
        if self.op in _LOGIC:
            return l.logical(r, self.op)
        if l.dtype.is_string() and self.op == "add":
            from ..kernels import strings as strk
            return strk.concat_str([l, r])
        from .. import kernels
        return kernels.binary_op(l, r, self.op)

    def __repr__(self):
        return f"({self.left!r} {self.op} {self.right!r})"


class Not(ExprNode):
    def __init__(self, child: ExprNode):
        self.child = child

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return Not(ch[0])

    def to_field(self, schema):
        return Field(self.child.to_field(schema).name, DataType.bool())

    def evaluate(self, batch) -> Series:
        return self.child.evaluate(batch).logical_not()

    def __repr__(self):
        return f"~{self.child!r}"


class IsNull(ExprNode):
    def __init__(self, child: ExprNode, negate: bool = False):
        self.child = child
        self.negate = negate

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return IsNull(ch[0], self.negate)

    def to_field(self, schema):
        return Field(self.child.to_field(schema).name, DataType.bool())

    def evaluate(self, batch) -> Series:
        s = self.child.evaluate(batch)
        return s.not_null() if self.negate else s.is_null()

    def __repr__(self):
        return f"{self.child!r}.{'not_null' if self.negate else 'is_null'}()"


class FillNull(ExprNode):
    def __init__(self, child: ExprNode, fill: ExprNode):
        self.child = child
        self.fill = fill

    def children(self):
        return [self.child, self.fill]

    def with_children(self, ch):
        return FillNull(ch[0], ch[1])

    def to_field(self, schema):
        f = self.child.to_field(schema)
        return Field(f.name, supertype(f.dtype, self.fill.to_field(schema).dtype))

    def evaluate(self, batch) -> Series:
        s = self.child.evaluate(batch)
        f = self.fill.evaluate(batch)
        if len(f) == 1:
            f = f.broadcast(len(s))
        return s.fill_null(f)

    def __repr__(self):
        return f"{self.child!r}.fill_null({self.fill!r})"


class Cast(ExprNode):
    def __init__(self, child: ExprNode, dtype: DataType):
        self.child = child
        self.dtype = dtype

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return Cast(ch[0], self.dtype)

    def to_field(self, schema):
        return Field(self.child.to_field(schema).name, self.dtype)

    def evaluate(self, batch) -> Series:
        return self.child.evaluate(batch).cast(self.dtype)

    def __repr__(self):
        return f"{self.child!r}.cast({self.dtype!r})"


class IsIn(ExprNode):
    def __init__(self, child: ExprNode, values: List[Any]):
        self.child = child
        self.values = values

    def children(self):
        return [self.child]

    def with_children(self, ch):
        return IsIn(ch[0], self.values)

    def to_field(self, schema):
        return Field(self.child.to_field(schema).name, DataType.bool())

    def evaluate(self, batch) -> Series:
        s = self.child.evaluate(batch)
        vals = Series.from_pylist("values", list(self.values),
                                  device=s.device)
        return s.is_in(vals)

    def __repr__(self):
        return f"{self.child!r}.is_in({self.values!r})"


class Between(ExprNode):
    def __init__(self, child: ExprNode, lo: ExprNode, hi: ExprNode):
        self.child = child
        self.lo = lo
        self.hi = hi

    def children(self):
        return [self.child, self.lo, self.hi]

    def with_children(self, ch):
        return Between(ch[0], ch[1], ch[2])

    def to_field(self, schema):
        return Field(self.child.to_field(schema).name, DataType.bool())

    def evaluate(self, batch) -> Series:
        s = self.child.evaluate(batch)
        lo = self.lo.evaluate(batch)
        hi = self.hi.evaluate(batch)
        return s.between(lo, hi)

    def __repr__(self):
        return f"{self.child!r}.between({self.lo!r},{self.hi!r})"


class IfElse(ExprNode):
    def __init__(self, pred: ExprNode, truthy: ExprNode, falsy: ExprNode):
        self.pred = pred
        self.truthy = truthy
        self.falsy = falsy

    def children(self):
        return [self.pred, self.truthy, self.falsy]

    def with_children(self, ch):
        return IfElse(ch[0], ch[1], ch[2])

    def out_name(self):
        return self.truthy.out_name()

    def to_field(self, schema):
        t = self.truthy.to_field(schema)
        f = self.falsy.to_field(schema)
        return Field(t.name, supertype(t.dtype, f.dtype))

    def evaluate(self, batch) -> Series:
        p = self.pred.evaluate(batch)
        t = self.truthy.evaluate(batch)
        f = self.falsy.evaluate(batch)
        return p.if_else(t, f)

    def __repr__(self):
        return f"if({self.pred!r}, {self.truthy!r}, {self.falsy!r})"


class Coalesce(ExprNode):
    def __init__(self, args: List[ExprNode]):
        self.args = args

    def children(self):
        return list(self.args)

    def with_children(self, ch):
        return Coalesce(ch)

    def to_field(self, schema):
        f0 = self.args[0].to_field(schema)
        dt = f0.dtype
        for a in self.args[1:]:
            dt = supertype(dt, a.to_field(schema).dtype)
        return Field(f0.name, dt)

    def evaluate(self, batch) -> Series:
        out = self.args[0].evaluate(batch)
        for a in self.args[1:]:
            nxt = a.evaluate(batch)
            if len(nxt) == 1:
                nxt = nxt.broadcast(len(out))
            out = out.fill_null(nxt)
        return out

    def __repr__(self):
        return f"coalesce({', '.join(map(repr, self.args))})"


class AggKind:
    SUM = "sum"
    MEAN = "mean"
    MIN = "min"
    MAX = "max"
    COUNT = "count"           # non-null count
    COUNT_ALL = "count_all"   # row count
    COUNT_DISTINCT = "count_distinct"
    ANY_VALUE = "any_value"
    LIST = "list"
    CONCAT = "concat"
    STDDEV = "stddev"
    VARIANCE = "variance"
    SKEW = "skew"
    APPROX_COUNT_DISTINCT = "approx_count_distinct"
    APPROX_PERCENTILE = "approx_percentile"
    BOOL_AND = "bool_and"
    BOOL_OR = "bool_or"
    # internal: DDSketch partial/final pair for distributed
    # approx_percentile (physical/sketch.py)
    SKETCH = "__sketch"
    SKETCH_FINAL = "__sketch_final"
    # user-defined aggregation (daft_amd.udaf): param = (instance, dtype)
    PY_UDAF = "__py_udaf"


_NUMERIC_AGGS = {AggKind.SUM, AggKind.MEAN, AggKind.STDDEV, AggKind.VARIANCE,
                 AggKind.SKEW}


class Agg(ExprNode):
    """Aggregation expression; evaluated by the aggregate operators."""

    def __init__(self, kind: str, child: Optional[ExprNode],
                 param: Any = None):
        self.kind = kind
        self.child = child
        self.param = param

    def children(self):
        return [self.child] if self.child is not None else []

    def with_children(self, ch):
        return Agg(self.kind, ch[0] if ch else None, self.param)

    def out_name(self):
        if self.child is None:
            return "count"
        return self.child.out_name()

    def is_aggregation(self):
        return True

    def to_field(self, schema: Schema) -> Field:
        if self.child is None:
            return Field("count", DataType.uint64())
        f = self.child.to_field(schema)
        k = self.kind
        if k in (AggKind.COUNT, AggKind.COUNT_ALL, AggKind.COUNT_DISTINCT,
                 AggKind.APPROX_COUNT_DISTINCT):
            return Field(f.name, DataType.uint64())
        if k == AggKind.SUM:
            if f.dtype.is_decimal():
                # widen precision for the running sum (ref: decimal sum
                # gets p=38).  p<=18 keeps int64 storage; wider inputs
                # sum exactly on two limbs (kernels/decimal128.py)
                return Field(f.name, DataType.decimal128(
                    18 if f.dtype.precision <= 18 else 38, f.dtype.scale))
            if f.dtype.is_integer():
                return Field(f.name, DataType.int64()
                             if f.dtype.is_signed_integer()
                             else DataType.uint64())
            return Field(f.name, DataType.float64()
                         if not f.dtype.is_floating() or
                         f.dtype.kind == TypeKind.FLOAT64
                         else DataType.float32())
        if k in (AggKind.MEAN, AggKind.STDDEV, AggKind.VARIANCE, AggKind.SKEW,
                 AggKind.APPROX_PERCENTILE):
            return Field(f.name, DataType.float64())
        if k in (AggKind.MIN, AggKind.MAX, AggKind.ANY_VALUE):
            return Field(f.name, f.dtype)
        if k in (AggKind.LIST, AggKind.CONCAT):
            return Field(f.name, DataType.list(f.dtype))
        if k in (AggKind.BOOL_AND, AggKind.BOOL_OR):
            return Field(f.name, DataType.bool())
        if k == AggKind.SKETCH:
            from ..physical.sketch import SKETCH_DTYPE
            return Field(f.name, SKETCH_DTYPE)
        if k == AggKind.SKETCH_FINAL:
            return Field(f.name, DataType.float64())
        if k == AggKind.PY_UDAF:
            return Field(f.name, self.param[1])
        raise ValueError(f"unknown agg kind {k}")

    def evaluate(self, batch) -> Series:
        raise RuntimeError(
            "aggregation expressions must run under an aggregate operator")

    def __repr__(self):
        return f"{self.child!r}.{self.kind}()" if self.child is not None \
            else "count(*)"


class ScalarFn(ExprNode):
    """Named scalar function with a vectorized Series implementation."""

    def __init__(self, name: str, fn: Callable[..., Series],
                 args: List[ExprNode], ret_dtype,
                 literal_args: Tuple = (), kwargs: Optional[dict] = None):
        self.name = name
        self.fn = fn
        self.args = args
        self.ret_dtype = ret_dtype  # DataType or callable(list[Field])->DataType
        self.literal_args = literal_args
        self.kwargs = kwargs or {}

    def children(self):
        return list(self.args)

    def with_children(self, ch):
        return ScalarFn(self.name, self.fn, ch, self.ret_dtype,
                        self.literal_args, self.kwargs)

    def to_field(self, schema):
        fields = [a.to_field(schema) for a in self.args]
        name = fields[0].name if fields else self.name
        dt = self.ret_dtype(fields) if callable(self.ret_dtype) \
            else self.ret_dtype
        return Field(name, dt)

    def evaluate(self, batch) -> Series:
        series = [a.evaluate(batch) for a in self.args]
        return self.fn(*series, *self.literal_args, **self.kwargs)

    def __repr__(self):
        inner = ", ".join(map(repr, self.args))
        extra = "".join(f", {a!r}" for a in self.literal_args)
        return f"{self.name}({inner}{extra})"


class PyUDF(ExprNode):
    """Row/batch-wise python UDF (ref: daft-dsl/src/python_udf/)."""

    def __init__(self, name: str, fn: Callable, args: List[ExprNode],
                 return_dtype: DataType, batched: bool = False,
                 max_retries: int = 0, on_error: str = "raise",
                 use_process: bool = False, concurrency: Optional[int] = None,
                 gpus: int = 0):
        self.name = name
        self.fn = fn
        self.args = args
        self.return_dtype = return_dtype
        self.batched = batched
        self.max_retries = max_retries
        self.on_error = on_error
        self.use_process = use_process
        self.concurrency = concurrency
        self.gpus = gpus

    def children(self):
        return list(self.args)

    def with_children(self, ch):
        return PyUDF(self.name, self.fn, ch, self.return_dtype, self.batched,
                     self.max_retries, self.on_error, self.use_process,
                     self.concurrency, self.gpus)

    def out_name(self):
        return self.name

    def to_field(self, schema):
        return Field(self.name, self.return_dtype)

    def evaluate(self, batch) -> Series:
        from ..udf import run_udf_node
        return run_udf_node(self, batch)

    def __repr__(self):
        return f"udf:{self.name}({', '.join(map(repr, self.args))})"


# ---------------------------------------------------------------------------
# user-facing Expression wrapper
# ---------------------------------------------------------------------------

def _to_node(v) -> ExprNode:
    if isinstance(v, Expression):
        return v._node
    if isinstance(v, ExprNode):
        return v
    return Literal(v)


class Expression:
    __slots__ = ("_node",)

    def __init__(self, node: ExprNode):
        self._node = node

    # naming / casting
    def alias(self, name: str) -> "Expression":
        return Expression(Alias(self._node, name))

    def cast(self, dtype: DataType) -> "Expression":
        return Expression(Cast(self._node, dtype))

    def name(self) -> str:
        return self._node.out_name()

    # arithmetic
    def __add__(self, o): return Expression(BinaryOp("add", self._node, _to_node(o)))
    def __radd__(self, o): return Expression(BinaryOp("add", _to_node(o), self._node))
    def __sub__(self, o): return Expression(BinaryOp("sub", self._node, _to_node(o)))
    def __rsub__(self, o): return Expression(BinaryOp("sub", _to_node(o), self._node))
    def __mul__(self, o): return Expression(BinaryOp("mul", self._node, _to_node(o)))
    def __rmul__(self, o): return Expression(BinaryOp("mul", _to_node(o), self._node))
    def __truediv__(self, o): return Expression(BinaryOp("div", self._node, _to_node(o)))
    def __rtruediv__(self, o): return Expression(BinaryOp("div", _to_node(o), self._node))
    def __floordiv__(self, o): return Expression(BinaryOp("floordiv", self._node, _to_node(o)))
    def __mod__(self, o): return Expression(BinaryOp("mod", self._node, _to_node(o)))
    def __pow__(self, o): return Expression(BinaryOp("pow", self._node, _to_node(o)))
    def __neg__(self): return Expression(BinaryOp("sub", Literal(0), self._node))

    # comparison
    def __eq__(self, o): return Expression(BinaryOp("eq", self._node, _to_node(o)))  # type: ignore
    def __ne__(self, o): return Expression(BinaryOp("ne", self._node, _to_node(o)))  # type: ignore
    def __lt__(self, o): return Expression(BinaryOp("lt", self._node, _to_node(o)))
    def __le__(self, o): return Expression(BinaryOp("le", self._node, _to_node(o)))
    def __gt__(self, o): return Expression(BinaryOp("gt", self._node, _to_node(o)))
    def __ge__(self, o): return Expression(BinaryOp("ge", self._node, _to_node(o)))

    def eq(self, o): return self.__eq__(o)
    def ne(self, o): return self.__ne__(o)

    def __hash__(self):
        return hash(repr(self._node))

    # logic
    def __and__(self, o): return Expression(BinaryOp("and", self._node, _to_node(o)))
    def __rand__(self, o): return Expression(BinaryOp("and", _to_node(o), self._node))
    def __or__(self, o): return Expression(BinaryOp("or", self._node, _to_node(o)))
    def __ror__(self, o): return Expression(BinaryOp("or", _to_node(o), self._node))
    def __xor__(self, o): return Expression(BinaryOp("xor", self._node, _to_node(o)))
    def __invert__(self): return Expression(Not(self._node))

    # null handling
    def is_null(self): return Expression(IsNull(self._node))
    def not_null(self): return Expression(IsNull(self._node, negate=True))
    def fill_null(self, fill): return Expression(FillNull(self._node, _to_node(fill)))

    def is_in(self, values) -> "Expression":
        if isinstance(values, Expression):
            raise TypeError("is_in expects a python list of literals")
        return Expression(IsIn(self._node, list(values)))

    def between(self, lo, hi) -> "Expression":
        return Expression(Between(self._node, _to_node(lo), _to_node(hi)))

    def if_else(self, truthy, falsy) -> "Expression":
        return Expression(IfElse(self._node, _to_node(truthy), _to_node(falsy)))

    def apply(self, fn: Callable, return_dtype: DataType) -> "Expression":
        return Expression(PyUDF(getattr(fn, "__name__", "apply"), fn,
                                [self._node], return_dtype))

    # aggregations
    def sum(self): return Expression(Agg(AggKind.SUM, self._node))
    def mean(self): return Expression(Agg(AggKind.MEAN, self._node))
    def avg(self): return self.mean()
    def min(self): return Expression(Agg(AggKind.MIN, self._node))
    def max(self): return Expression(Agg(AggKind.MAX, self._node))
    def count(self, mode: str = "valid"):
        kind = AggKind.COUNT_ALL if mode == "all" else AggKind.COUNT
        return Expression(Agg(kind, self._node))
    def count_distinct(self): return Expression(Agg(AggKind.COUNT_DISTINCT, self._node))
    def approx_count_distinct(self):
        return Expression(Agg(AggKind.APPROX_COUNT_DISTINCT, self._node))
    def approx_percentile(self, q: float):
        return Expression(Agg(AggKind.APPROX_PERCENTILE, self._node, q))
    def any_value(self): return Expression(Agg(AggKind.ANY_VALUE, self._node))
    def agg_list(self): return Expression(Agg(AggKind.LIST, self._node))
    def agg_concat(self): return Expression(Agg(AggKind.CONCAT, self._node))
    def stddev(self): return Expression(Agg(AggKind.STDDEV, self._node))
    def variance(self): return Expression(Agg(AggKind.VARIANCE, self._node))
    def skew(self): return Expression(Agg(AggKind.SKEW, self._node))
    def bool_and(self): return Expression(Agg(AggKind.BOOL_AND, self._node))
    def bool_or(self): return Expression(Agg(AggKind.BOOL_OR, self._node))

    # window
    def over(self, window) -> "Expression":
        from ..physical.window import WindowFn
        node = self._node
        base = node.child if isinstance(node, Alias) else node
        name = node.out_name()
        if isinstance(base, WindowFn):
            wf = WindowFn(base.kind, base.inner, window, base.offset,
                          base.default)
        elif isinstance(base, Agg):
            wf = WindowFn("agg", base, window)
        else:
            raise TypeError(
                "over() expects an aggregation expression (or use "
                "daft_amd.functions.row_number/rank/lag/lead)")
        return Expression(Alias(wf, name) if isinstance(node, Alias) else wf)

    def lag(self, offset: int = 1, default=None) -> "Expression":
        """Use together with .over(window)."""
        from ..physical.window import WindowFn
        return Expression(WindowFn("lag", self._node, None, offset, default))

    def lead(self, offset: int = 1, default=None) -> "Expression":
        from ..physical.window import WindowFn
        return Expression(WindowFn("lead", self._node, None, offset, default))

    # namespaces
    @property
    def str(self) -> "StringNamespace":
        return StringNamespace(self._node)

    @property
    def dt(self) -> "TemporalNamespace":
        return TemporalNamespace(self._node)

    @property
    def list(self) -> "ListNamespace":
        return ListNamespace(self._node)

    @property
    def struct(self) -> "StructNamespace":
        return StructNamespace(self._node)

    @property
    def float(self) -> "FloatNamespace":
        return FloatNamespace(self._node)

    @property
    def embedding(self) -> "EmbeddingNamespace":
        return EmbeddingNamespace(self._node)

    @property
    def image(self) -> "ImageNamespace":
        from ..functions.image import ImageNamespace
        return ImageNamespace(self._node)

    @property
    def url(self) -> "UrlNamespace":
        return UrlNamespace(self._node)

    @property
    def binary(self) -> "BinaryNamespace":
        return BinaryNamespace(self._node)

    @property
    def map(self) -> "MapNamespace":
        return MapNamespace(self._node)

    @property
    def partitioning(self) -> "PartitioningNamespace":
        return PartitioningNamespace(self._node)

    # ---- method-form conveniences mirroring the reference Expression
    # surface; each delegates to the equivalent daft_amd.functions entry
    # (ref: daft/expressions/expressions.py method list) ----------------
    def explode(self):
        from ..functions.aliases import _Explode
        return _Explode(self)

    def unnest(self):
        from ..functions.aliases import _Unnest
        return _Unnest(self)

    def eq_null_safe(self, other) -> "Expression":
        from ..functions import eq_null_safe as _f
        return _f(self, other)

    def try_cast(self, dtype) -> "Expression":
        from ..functions import try_cast as _f
        return _f(self, dtype)

    def median(self) -> "Expression":
        from ..functions import median as _f
        return _f(self)

    def fill_nan(self, value) -> "Expression":
        from ..functions import fill_nan as _f
        return _f(self, value)

    def is_nan(self) -> "Expression":
        return self.float.is_nan()

    def not_nan(self) -> "Expression":
        return ~self.float.is_nan()

    def is_inf(self) -> "Expression":
        return self.float.is_inf()

    def shift_left(self, other) -> "Expression":
        from ..functions import shift_left as _f
        return _f(self, other)

    def shift_right(self, other) -> "Expression":
        from ..functions import shift_right as _f
        return _f(self, other)

    def bitwise_and(self, other) -> "Expression":
        from ..functions import bitwise_and as _f
        return _f(self, other)

    def bitwise_or(self, other) -> "Expression":
        from ..functions import bitwise_or as _f
        return _f(self, other)

    def bitwise_xor(self, other) -> "Expression":
        from ..functions import bitwise_xor as _f
        return _f(self, other)

    def hash(self, seed: int = 0) -> "Expression":
        from .. import kernels

        def _h(s: Series, sd) -> Series:
            h = kernels.hash_columns([s], sd)
            return Series(s.name, DataType.uint64(), data=h.view(torch.uint64))
        return Expression(ScalarFn("hash", _h, [self._node],
                                   DataType.uint64(), (seed,)))

    def minhash(self, num_hashes: int, ngram_size: int = 1,
                seed: int = 1) -> "Expression":
        from ..functions.minhash import minhash_series
        return Expression(ScalarFn(
            "minhash", minhash_series, [self._node],
            DataType.fixed_size_list(DataType.uint32(), num_hashes),
            (num_hashes, ngram_size, seed)))

    def simhash(self, ngram_size: int = 4) -> "Expression":
        from ..functions.minhash import simhash_series
        return Expression(ScalarFn(
            "simhash", simhash_series, [self._node],
            DataType.uint64(), (ngram_size,)))

    # misc
    def abs(self):
        return Expression(ScalarFn(
            "abs", _series_abs, [self._node],
            lambda f: f[0].dtype))

    def round(self, decimals: int = 0):
        return Expression(ScalarFn(
            "round", _series_round, [self._node],
            lambda f: f[0].dtype, (decimals,)))

    def floor(self):
        return Expression(ScalarFn("floor", _series_floor, [self._node],
                                   lambda f: f[0].dtype))

    def ceil(self):
        return Expression(ScalarFn("ceil", _series_ceil, [self._node],
                                   lambda f: f[0].dtype))

    def sqrt(self):
        return Expression(ScalarFn("sqrt", _series_sqrt, [self._node],
                                   DataType.float64()))

    def exp(self):
        return Expression(ScalarFn("exp", _series_exp, [self._node],
                                   DataType.float64()))

    def log(self, base: Optional[float] = None):
        return Expression(ScalarFn("log", _series_log, [self._node],
                                   DataType.float64(), (base,)))

    def clip(self, lo=None, hi=None):
        return Expression(ScalarFn("clip", _series_clip, [self._node],
                                   lambda f: f[0].dtype, (lo, hi)))

    def __repr__(self):
        return repr(self._node)


def _series_unary(fn):
    def impl(s: Series, *args) -> Series:
        out = fn(s.data, *args)
        return Series(s.name, s.dtype if not out.dtype.is_floating_point or
                      s.dtype.is_floating() else DataType.float64(),
                      data=out, validity=s.validity)
    return impl


def _series_abs(s: Series) -> Series:
    return Series(s.name, s.dtype, data=torch.abs(s.data),
                  validity=s.validity)


def _series_round(s: Series, decimals: int) -> Series:
    if s.dtype.is_integer():
        return s
    return Series(s.name, s.dtype, data=torch.round(s.data, decimals=decimals),
                  validity=s.validity)


def _series_floor(s: Series) -> Series:
    if s.dtype.is_integer():
        return s
    return Series(s.name, s.dtype, data=torch.floor(s.data),
                  validity=s.validity)


def _series_ceil(s: Series) -> Series:
    if s.dtype.is_integer():
        return s
    return Series(s.name, s.dtype, data=torch.ceil(s.data),
                  validity=s.validity)


def _series_sqrt(s: Series) -> Series:
    return Series(s.name, DataType.float64(),
                  data=torch.sqrt(s.data.to(torch.float64)),
                  validity=s.validity)


def _series_exp(s: Series) -> Series:
    return Series(s.name, DataType.float64(),
                  data=torch.exp(s.data.to(torch.float64)),
                  validity=s.validity)


def _series_log(s: Series, base) -> Series:
    d = torch.log(s.data.to(torch.float64))
    if base is not None:
        d = d / torch.log(torch.tensor(float(base)))
    return Series(s.name, DataType.float64(), data=d, validity=s.validity)


def _series_clip(s: Series, lo, hi) -> Series:
    return Series(s.name, s.dtype, data=torch.clamp(s.data, lo, hi),
                  validity=s.validity)


# ---------------------------------------------------------------------------
# namespaces
# ---------------------------------------------------------------------------

class _Namespace:
    def __init__(self, node: ExprNode):
        self._node = node

    def _fn(self, name, fn, ret, *literal_args, **kwargs) -> Expression:
        return Expression(ScalarFn(name, fn, [self._node], ret,
                                   tuple(literal_args), kwargs))


class StringNamespace(_Namespace):
    def contains(self, pat: str):
        from ..kernels import strings as k
        return self._fn("contains", k.contains, DataType.bool(), pat)

    def startswith(self, pat: str):
        from ..kernels import strings as k
        return self._fn("startswith", k.startswith, DataType.bool(), pat)

    def endswith(self, pat: str):
        from ..kernels import strings as k
        return self._fn("endswith", k.endswith, DataType.bool(), pat)

    def like(self, pattern: str):
        from ..kernels import strings as k
        return self._fn("like", k.like, DataType.bool(), pattern)

    def ilike(self, pattern: str):
        from ..kernels import strings as k
        return self._fn("ilike", k.like, DataType.bool(), pattern, True)

    def match(self, pattern: str):
        from ..kernels import strings as k
        return self._fn("regexp_match", k.regexp_match, DataType.bool(), pattern)

    def length(self):
        from ..kernels import strings as k
        return self._fn("length", k.length, DataType.uint64())

    def length_bytes(self):
        from ..kernels import strings as k
        return self._fn("length_bytes", k.length_bytes, DataType.uint64())

    def lower(self):
        from ..kernels import strings as k
        return self._fn("lower", k.lower, DataType.string())

    def upper(self):
        from ..kernels import strings as k
        return self._fn("upper", k.upper, DataType.string())

    def lstrip(self):
        from ..kernels import strings as k
        return self._fn("lstrip", k.lstrip, DataType.string())

    def rstrip(self):
        from ..kernels import strings as k
        return self._fn("rstrip", k.rstrip, DataType.string())

    def strip(self):
        from ..kernels import strings as k
        return self._fn("strip", k.strip, DataType.string())

    def reverse(self):
        from ..kernels import strings as k
        return self._fn("reverse", k.reverse, DataType.string())

    def capitalize(self):
        from ..kernels import strings as k
        return self._fn("capitalize", k.capitalize, DataType.string())

    def substr(self, start: int, length: Optional[int] = None):
        from ..kernels import strings as k
        return self._fn("substr", k.substr, DataType.string(), start, length)

    def left(self, n: int):
        from ..kernels import strings as k
        return self._fn("left", k.left, DataType.string(), n)

    def right(self, n: int):
        from ..kernels import strings as k
        return self._fn("right", k.right, DataType.string(), n)

    def find(self, pat: str):
        from ..kernels import strings as k
        return self._fn("find", k.find, DataType.int64(), pat)

    def split(self, sep: str):
        from ..kernels import strings as k
        return self._fn("split", k.split, DataType.list(DataType.string()), sep)

    def concat(self, other):
        from ..kernels import strings as k
        return Expression(ScalarFn(
            "concat", lambda a, b: k.concat_str([a, b]),
            [self._node, _to_node(other)], DataType.string()))

    def repeat(self, n: int):
        from ..kernels import strings as k
        return self._fn("repeat", k.repeat, DataType.string(), n)

    def lpad(self, width: int, fillchar: str = " "):
        from ..kernels import strings as k
        return self._fn("lpad", k.lpad, DataType.string(), width, fillchar)

    def rpad(self, width: int, fillchar: str = " "):
        from ..kernels import strings as k
        return self._fn("rpad", k.rpad, DataType.string(), width, fillchar)

    def to_date(self, fmt: str = "%Y-%m-%d"):
        return self._fn("to_date", _str_to_date, DataType.date(), fmt)

    def tokenize_encode(self, tokenizer: str = "simple"):
        from ..functions.tokenize import tokenize_encode_series
        return self._fn("tokenize_encode", tokenize_encode_series,
                        DataType.list(DataType.int32()), tokenizer)

    def tokenize_decode(self, tokenizer: str = "simple"):
        from ..functions.tokenize import tokenize_decode_series
        return self._fn("tokenize_decode", tokenize_decode_series,
                        DataType.string(), tokenizer)

    # ---- method forms delegating to daft_amd.functions (ref:
    # ExpressionStringNamespace surface) --------------------------------
    def _delegate(self, fname, *args, **kwargs):
        from .. import functions as F
        return getattr(F, fname)(Expression(self._node), *args, **kwargs)

    def normalize(self, **kwargs):
        return self._delegate("normalize", **kwargs)

    def count_matches(self, patterns, whole_words: bool = False,
                      case_sensitive: bool = True):
        return self._delegate("count_matches", patterns, whole_words,
                              case_sensitive)

    def replace(self, search: str, replacement: str, regex: bool = False):
        if regex:
            return self._delegate("regexp_replace", search, replacement)
        import re as _re
        return self._delegate("regexp_replace", _re.escape(search),
                              replacement)

    def regexp_replace(self, pattern: str, replacement: str):
        return self._delegate("regexp_replace", pattern, replacement)

    def extract(self, pattern: str, group: int = 0):
        return self._delegate("regexp_extract", pattern, group)

    def extract_all(self, pattern: str, group: int = 0):
        return self._delegate("regexp_extract_all", pattern, group)

    def regexp_split(self, pattern: str):
        return self._delegate("regexp_split", pattern)

    def split_part(self, delim: str, n: int):
        return self._delegate("split_part", delim, n)

    def substring_index(self, delim: str, n: int):
        return self._delegate("substring_index", delim, n)

    def translate(self, src: str, dst: str):
        return self._delegate("translate", src, dst)

    def to_snake_case(self):
        return self._delegate("to_snake_case")

    def to_camel_case(self):
        return self._delegate("to_camel_case")

    def to_kebab_case(self):
        return self._delegate("to_kebab_case")

    def to_title_case(self):
        return self._delegate("to_title_case")

    def levenshtein(self, other):
        from ..functions import levenshtein_distance as _f
        return _f(Expression(self._node), other)

    def jaro_winkler(self, other):
        from ..functions import jaro_winkler_similarity as _f
        return _f(Expression(self._node), other)


def _str_to_date(s: Series, fmt: str) -> Series:
    vals = s.cpu().to_pylist()
    out = [None if v is None else _dt.datetime.strptime(v, fmt).date()
           for v in vals]
    res = Series.from_pylist(s.name, out, DataType.date())
    return res.to(s.device) if s.is_gpu() else res


class TemporalNamespace(_Namespace):
    def _civil(self, part: str) -> Expression:
        return self._fn(part, _dt_extract, DataType.int32(), part)

    def year(self): return self._civil("year")
    def month(self): return self._civil("month")
    def day(self): return self._civil("day")
    def quarter(self): return self._civil("quarter")
    def day_of_week(self): return self._civil("day_of_week")
    def day_of_year(self): return self._civil("day_of_year")
    def week_of_year(self): return self._civil("week_of_year")
    def hour(self): return self._civil("hour")
    def minute(self): return self._civil("minute")
    def second(self): return self._civil("second")

    def date(self):
        return self._fn("date", _dt_to_date, DataType.date())

    def truncate(self, interval: str):
        return self._fn("truncate", _dt_truncate, lambda f: f[0].dtype,
                        interval)

    def total_seconds(self):
        return self._fn("total_seconds", _dur_total, DataType.int64(),
                        1_000_000)

    def total_days(self):
        return self._fn("total_days", _dur_total, DataType.int64(),
                        86_400_000_000)

    def total_hours(self):
        return self._fn("total_hours", _dur_total, DataType.int64(),
                        3_600_000_000)

    def total_minutes(self):
        return self._fn("total_minutes", _dur_total, DataType.int64(),
                        60_000_000)

    def total_milliseconds(self):
        return self._fn("total_milliseconds", _dur_total,
                        DataType.int64(), 1_000)

    def total_microseconds(self):
        return self._fn("total_microseconds", _dur_total,
                        DataType.int64(), 1)

    def total_nanoseconds(self):
        return self._fn("total_nanoseconds", _dur_total_ns,
                        DataType.int64())

    # method forms delegating to daft_amd.functions (ref:
    # ExpressionDatetimeNamespace)
    def _delegate(self, fname, *args, **kwargs):
        from .. import functions as F
        return getattr(F, fname)(Expression(self._node), *args, **kwargs)

    def strftime(self, fmt: str = "%Y-%m-%d"):
        return self._delegate("strftime", fmt)

    def to_unix_epoch(self, unit: str = "s"):
        return self._delegate("to_unix_epoch", unit)

    def unix_date(self):
        return self._delegate("unix_date")

    def time(self):
        return self._delegate("time")

    def day_of_month(self):
        return self._civil("day")

    def millisecond(self):
        return self._fn("millisecond", _dt_subsecond, DataType.int32(),
                        1_000, 1_000)

    def microsecond(self):
        return self._fn("microsecond", _dt_subsecond, DataType.int32(),
                        1, 1_000_000)

    def nanosecond(self):
        return self._fn("nanosecond", _dt_subsecond, DataType.int32(),
                        None, None)

    def date_trunc(self, interval: str):
        return self.truncate(interval)


def _days_to_civil(days: torch.Tensor):
    """Howard Hinnant's civil_from_days, vectorized on torch int ops."""
    z = days.to(torch.int64) + 719468
    era = torch.div(torch.where(z >= 0, z, z - 146096), 146097,
                    rounding_mode="floor")
    doe = z - era * 146097
    yoe = torch.div(doe - torch.div(doe, 1460, rounding_mode="floor")
                    + torch.div(doe, 36524, rounding_mode="floor")
                    - torch.div(doe, 146096, rounding_mode="floor"),
                    365, rounding_mode="floor")
    y = yoe + era * 400
    doy = doe - (365 * yoe + torch.div(yoe, 4, rounding_mode="floor")
                 - torch.div(yoe, 100, rounding_mode="floor"))
    mp = torch.div(5 * doy + 2, 153, rounding_mode="floor")
    d = doy - torch.div(153 * mp + 2, 5, rounding_mode="floor") + 1
    m = mp + torch.where(mp < 10, torch.full_like(mp, 3),
                         torch.full_like(mp, -9))
    y = y + (m <= 2).to(torch.int64)
    return y, m, d, doy


def _ts_to_days_and_us(s: Series):
    if s.dtype.kind == TypeKind.DATE:
        return s.data.to(torch.int64), None
    unit = s.dtype.timeunit
    mult = {"s": 1_000_000, "ms": 1_000, "us": 1, "ns": 1}[unit]
    us = s.data if unit != "ns" else torch.div(s.data, 1000,
                                               rounding_mode="floor")
    us = us * mult if mult != 1 else us
    days = torch.div(us, 86_400_000_000, rounding_mode="floor")
    tod = us - days * 86_400_000_000
    return days, tod


def _dt_extract(s: Series, part: str) -> Series:
    days, tod = _ts_to_days_and_us(s)
    if part in ("hour", "minute", "second"):
        assert tod is not None, f"{part} requires a timestamp"
        sec = torch.div(tod, 1_000_000, rounding_mode="floor")
        if part == "hour":
            out = torch.div(sec, 3600, rounding_mode="floor")
        elif part == "minute":
            out = torch.div(sec, 60, rounding_mode="floor") % 60
        else:
            out = sec % 60
        return Series(s.name, DataType.int32(), data=out.to(torch.int32),
                      validity=s.validity)
    y, m, d, _doy = _days_to_civil(days)
    if part == "year":
        out = y
    elif part == "month":
        out = m
    elif part == "day":
        out = d
    elif part == "quarter":
        out = torch.div(m - 1, 3, rounding_mode="floor") + 1
    elif part == "day_of_week":
        out = (days + 3) % 7  # 1970-01-01 was a Thursday; 0 = Monday
    elif part == "day_of_year":
        jan1 = _civil_to_days(y, torch.ones_like(m), torch.ones_like(d))
        out = days - jan1 + 1
    elif part == "week_of_year":
        jan1 = _civil_to_days(y, torch.ones_like(m), torch.ones_like(d))
        out = torch.div(days - jan1, 7, rounding_mode="floor") + 1
    else:
        raise ValueError(part)
    return Series(s.name, DataType.int32(), data=out.to(torch.int32),
                  validity=s.validity)


def _civil_to_days(y: torch.Tensor, m: torch.Tensor,
                   d: torch.Tensor) -> torch.Tensor:
    """days_from_civil, vectorized."""
    y = y - (m <= 2).to(torch.int64)
    era = torch.div(torch.where(y >= 0, y, y - 399), 400,
                    rounding_mode="floor")
    yoe = y - era * 400
    mp = torch.where(m > 2, m - 3, m + 9)
    doy = torch.div(153 * mp + 2, 5, rounding_mode="floor") + d - 1
    doe = yoe * 365 + torch.div(yoe, 4, rounding_mode="floor") \
        - torch.div(yoe, 100, rounding_mode="floor") + doy
    return era * 146097 + doe - 719468


def _dt_to_date(s: Series) -> Series:
    days, _ = _ts_to_days_and_us(s)
    return Series(s.name, DataType.date(), data=days.to(torch.int32),
                  validity=s.validity)


def _dt_truncate(s: Series, interval: str) -> Series:
    days, tod = _ts_to_days_and_us(s)
    y, m, d, _ = _days_to_civil(days)
    if interval in ("year", "1 year"):
        new_days = _civil_to_days(y, torch.ones_like(m), torch.ones_like(d))
    elif interval in ("month", "1 month"):
        new_days = _civil_to_days(y, m, torch.ones_like(d))
    elif interval in ("week", "1 week"):
        new_days = days - ((days + 3) % 7)
    elif interval in ("day", "1 day"):
        new_days = days
    else:
        # sub-day intervals ("hour", "15 minutes", "30 seconds", ...):
        # truncate the raw timestamp to the interval width
        parts = interval.split()
        count = 1
        unit_word = parts[-1].rstrip("s")
        if len(parts) == 2:
            try:
                count = int(parts[0])
            except ValueError:
                raise ValueError(
                    f"unsupported truncate interval {interval}")
        width_us = {"hour": 3_600_000_000, "minute": 60_000_000,
                    "second": 1_000_000, "millisecond": 1_000,
                    "microsecond": 1}.get(unit_word)
        if width_us is None or s.dtype.kind == TypeKind.DATE:
            raise ValueError(f"unsupported truncate interval {interval}")
        unit = s.dtype.timeunit
        mult = {"s": 1, "ms": 10**3, "us": 10**6, "ns": 10**9}[unit]
        width = width_us * count * mult // 10**6
        if width == 0:
            raise ValueError(f"interval below {unit} resolution: "
                             f"{interval}")
        out = torch.div(s.data, width, rounding_mode="floor") * width
        return Series(s.name, s.dtype, data=out, validity=s.validity)
    if s.dtype.kind == TypeKind.DATE:
        return Series(s.name, s.dtype, data=new_days.to(torch.int32),
                      validity=s.validity)
    unit = s.dtype.timeunit
    mult = {"s": 1, "ms": 10**3, "us": 10**6, "ns": 10**9}[unit]
    out = new_days * 86_400 * mult
    return Series(s.name, s.dtype, data=out, validity=s.validity)


def _dur_total(s: Series, div_us: int) -> Series:
    assert s.dtype.kind == TypeKind.DURATION
    mult = {"s": 1_000_000, "ms": 1_000, "us": 1, "ns": 1}[s.dtype.timeunit]
    us = s.data * mult if s.dtype.timeunit != "ns" else \
        torch.div(s.data, 1000, rounding_mode="floor")
    out = torch.div(us, div_us, rounding_mode="floor")
    return Series(s.name, DataType.int64(), data=out, validity=s.validity)


def _dur_total_ns(s: Series) -> Series:
    assert s.dtype.kind == TypeKind.DURATION
    mult = {"s": 1_000_000_000, "ms": 1_000_000, "us": 1_000, "ns": 1}[
        s.dtype.timeunit]
    return Series(s.name, DataType.int64(), data=s.data * mult,
                  validity=s.validity)


def _dt_subsecond(s: Series, div, mod) -> Series:
    """Sub-second components of a timestamp (pandas/arrow semantics:
    millisecond 0..999, microsecond 0..999999, nanosecond 0..999 within
    the microsecond)."""
    unit = s.dtype.timeunit
    if div is None:                     # nanosecond component
        if unit != "ns":
            out = torch.zeros_like(s.data, dtype=torch.int64)
        else:
            out = ((s.data % 1000) + 1000) % 1000
        return Series(s.name, DataType.int32(),
                      data=out.to(torch.int32), validity=s.validity)
    mult = {"s": 1_000_000, "ms": 1_000, "us": 1}.get(unit)
    us = s.data * mult if mult is not None else \
        torch.div(s.data, 1000, rounding_mode="floor")
    frac = ((us % 1_000_000) + 1_000_000) % 1_000_000
    out = torch.div(frac, div, rounding_mode="floor") % mod
    return Series(s.name, DataType.int32(), data=out.to(torch.int32),
                  validity=s.validity)


class ListNamespace(_Namespace):
    def length(self):
        return self._fn("list_length", _list_length, DataType.uint64())

    def _delegate(self, fname, *args, **kwargs):
        from .. import functions as F
        return getattr(F, fname)(Expression(self._node), *args, **kwargs)

    def append(self, value):
        return self._delegate("list_append", value)

    def filter(self, fn):
        return self._delegate("list_filter", fn)

    def map(self, fn):
        return self._delegate("list_map", fn)

    def count(self):
        return self._delegate("list_count")

    def bool_and(self):
        return self._delegate("list_bool_and")

    def bool_or(self):
        return self._delegate("list_bool_or")

    def get(self, idx: int, default=None):
        return self._fn("list_get", _list_get,
                        lambda f: f[0].dtype.inner, idx, default)

    def sum(self):
        return self._fn("list_sum", _list_agg,
                        lambda f: f[0].dtype.inner, "sum")

    def mean(self):
        return self._fn("list_mean", _list_agg, DataType.float64(), "mean")

    def min(self):
        return self._fn("list_min", _list_agg,
                        lambda f: f[0].dtype.inner, "min")

    def max(self):
        return self._fn("list_max", _list_agg,
                        lambda f: f[0].dtype.inner, "max")

    def join(self, sep: str):
        return self._fn("list_join", _list_join, DataType.string(), sep)

    def distinct(self):
        return self._fn("list_distinct", _list_distinct,
                        lambda f: f[0].dtype)

    def value_counts(self):
        def ret(fields):
            inner = fields[0].dtype.inner
            return DataType.list(DataType.struct(
                {"value": inner, "count": DataType.uint64()}))
        return self._fn("list_value_counts", _list_value_counts, ret)

    def contains(self, value):
        return self._fn("list_contains", _list_contains, DataType.bool(),
                        value)

    def chunk(self, size: int):
        def ret(fields):
            return DataType.list(DataType.list(fields[0].dtype.inner))
        return self._fn("list_chunk", _list_chunk, ret, size)

    def slice(self, start: int, end: Optional[int] = None):
        return self._fn("list_slice", _list_slice,
                        lambda f: f[0].dtype, start, end)

    def sort(self, desc: bool = False):
        return self._fn("list_sort", _list_sort,
                        lambda f: f[0].dtype, desc)

    def flatten(self):
        def ret(fields):
            return fields[0].dtype.inner
        return self._fn("list_flatten", _list_flatten, ret)

    def reverse(self):
        return self._fn("list_reverse", _list_reverse,
                        lambda f: f[0].dtype)


def _list_sort(s: Series, desc: bool) -> Series:
    from ..schema import DataType as DT

    def srt(v):
        nn = sorted((x for x in v if x is not None), reverse=desc)
        nulls = [None] * (len(v) - len(nn))
        return nn + nulls
    return _list_pylist_map(s, srt, DT.list(s.dtype.inner)
                            if s.dtype.kind == TypeKind.LIST else s.dtype)


def _list_flatten(s: Series) -> Series:
    from ..schema import DataType as DT

    def fl(v):
        out = []
        for x in v:
            if isinstance(x, list):
                out.extend(x)
            elif x is not None:
                out.append(x)
        return out
    inner = s.dtype.inner
    if inner is not None and inner.kind == TypeKind.LIST:
        inner = inner.inner
    return _list_pylist_map(s, fl, DT.list(inner))


def _list_reverse(s: Series) -> Series:
    from ..schema import DataType as DT
    return _list_pylist_map(s, lambda v: list(reversed(v)),
                            DT.list(s.dtype.inner)
                            if s.dtype.kind == TypeKind.LIST else s.dtype)


def _list_length(s: Series) -> Series:
    if s.dtype.kind == TypeKind.LIST:
        lens = s.offsets[1:] - s.offsets[:-1]
    else:
        lens = torch.full((len(s),), s.dtype.size, dtype=torch.int64,
                          device=s.device)
    return Series(s.name, DataType.uint64(), data=lens.view(torch.uint64),
                  validity=s.validity)


def _list_get(s: Series, idx: int, default) -> Series:
    if s.dtype.kind == TypeKind.LIST:
        lens = s.offsets[1:] - s.offsets[:-1]
        child_idx = torch.where(
            lens > idx, s.offsets[:-1] + idx,
            torch.full_like(lens, -1))
    else:
        n = s.dtype.size
        base = torch.arange(len(s), device=s.device, dtype=torch.int64) * n
        child_idx = base + idx if idx < n else torch.full_like(base, -1)
    out = s.children[0].take(child_idx)
    return out.rename(s.name)


def _list_agg(s: Series, op: str) -> Series:
    from . import expressions as _  # noqa
    from ..kernels import rowops
    if s.dtype.kind == TypeKind.LIST:
        n = len(s)
        # group child elements by parent row
        lens = (s.offsets[1:] - s.offsets[:-1])
        gid = torch.repeat_interleave(
            torch.arange(n, device=s.device, dtype=torch.int64), lens)
        kind = "sum" if op in ("sum", "mean") else op
        child = s.children[0]
        data, cnt = rowops.grouped_agg(gid, n, child, kind)
        if op == "mean":
            out = data.to(torch.float64) / cnt.clamp(min=1).to(torch.float64)
            validity = cnt > 0
            return Series(s.name, DataType.float64(), data=out,
                          validity=validity)
        validity = cnt > 0
        dt = s.dtype.inner if op != "sum" else (
            DataType.int64() if s.dtype.inner.is_integer()
            else DataType.float64())
        return Series(s.name, dt, data=data.to(dt.to_torch()),
                      validity=validity)
    # fixed size list: tensor reshape reduction
    n = len(s)
    sz = s.dtype.size
    mat = s.children[0].data.reshape(n, sz)
    if op == "sum":
        out = mat.sum(dim=1)
    elif op == "mean":
        out = mat.to(torch.float64).mean(dim=1)
    elif op == "min":
        out = mat.min(dim=1).values
    else:
        out = mat.max(dim=1).values
    from ..schema import from_torch_dtype
    return Series(s.name, from_torch_dtype(out.dtype), data=out,
                  validity=s.validity)


def _list_join(s: Series, sep: str) -> Series:
    vals = s.cpu().to_pylist()
    out = [None if v is None else sep.join("" if x is None else str(x)
                                           for x in v) for v in vals]
    res = Series.from_pylist(s.name, out, DataType.string())
    return res.to(s.device) if s.is_gpu() else res


def _list_pylist_map(s: Series, f, out_dtype) -> Series:
    vals = s.cpu().to_pylist()
    out = [None if v is None else f(v) for v in vals]
    res = Series.from_pylist(s.name, out, out_dtype)
    return res.to(s.device) if s.is_gpu() else res


def _list_distinct(s: Series) -> Series:
    from ..schema import DataType as DT

    def dedup(v):
        seen, out = set(), []
        for x in v:
            k = repr(x)
            if k not in seen:
                seen.add(k)
                out.append(x)
        return out
    return _list_pylist_map(s, dedup, DT.list(s.dtype.inner)
                            if s.dtype.kind == TypeKind.LIST else s.dtype)


def _list_value_counts(s: Series) -> Series:
    from collections import Counter
    from ..schema import DataType as DT
    inner = s.dtype.inner

    def vc(v):
        c = Counter(x for x in v if x is not None)
        return [{"value": k, "count": n} for k, n in c.items()]
    out_dt = DT.list(DT.struct({"value": inner, "count": DT.uint64()}))
    return _list_pylist_map(s, vc, out_dt)


def _list_contains(s: Series, value) -> Series:
    vals = s.cpu().to_pylist()
    out = [None if v is None else (value in v) for v in vals]
    res = Series.from_pylist(s.name, out, DataType.bool())
    return res.to(s.device) if s.is_gpu() else res


def _list_chunk(s: Series, size: int) -> Series:
    from ..schema import DataType as DT

    def ch(v):
        return [v[i:i + size] for i in range(0, len(v), size)]
    return _list_pylist_map(s, ch, DT.list(DT.list(s.dtype.inner)))


def _list_slice(s: Series, start: int, end) -> Series:
    from ..schema import DataType as DT

    def sl(v):
        return v[start:end]
    dt = s.dtype if s.dtype.kind == TypeKind.LIST else DT.list(s.dtype.inner)
    return _list_pylist_map(s, sl, dt)


class StructNamespace(_Namespace):
    def get(self, field_name: str):
        return self._fn("struct_get", _struct_get,
                        lambda f: next(x.dtype for x in f[0].dtype.fields
                                       if x.name == field_name), field_name)


def _struct_get(s: Series, field_name: str) -> Series:
    for i, f in enumerate(s.dtype.fields):
        if f.name == field_name:
            child = s.children[i]
            if s.validity is not None:
                v = child.validity & s.validity if child.validity is not None \
                    else s.validity.clone()
                child = child.with_validity(v)
            return child.rename(field_name)
    raise KeyError(field_name)


class MapNamespace(_Namespace):
    """col.map.get/keys/values over the first-class Map type (ref:
    ExpressionMapNamespace, daft/expressions/expressions.py)."""

    def get(self, key):
        from ..functions import map_get as _f
        return _f(Expression(self._node), key)

    def keys(self):
        from ..functions import map_keys as _f
        return _f(Expression(self._node))

    def values(self):
        from ..functions import map_values as _f
        return _f(Expression(self._node))


class PartitioningNamespace(_Namespace):
    """Iceberg-style partition transforms (ref:
    ExpressionPartitioningNamespace)."""

    def days(self):
        from ..functions import partition_days as _f
        return _f(Expression(self._node))

    def hours(self):
        from ..functions import partition_hours as _f
        return _f(Expression(self._node))

    def months(self):
        from ..functions import partition_months as _f
        return _f(Expression(self._node))

    def years(self):
        from ..functions import partition_years as _f
        return _f(Expression(self._node))

    def iceberg_bucket(self, n: int):
        from ..functions import partition_iceberg_bucket as _f
        return _f(Expression(self._node), n)

    def iceberg_truncate(self, w):
        from ..functions import partition_iceberg_truncate as _f
        return _f(Expression(self._node), w)


class FloatNamespace(_Namespace):
    def is_nan(self):
        return self._fn("is_nan", _float_is_nan, DataType.bool())

    def is_inf(self):
        return self._fn("is_inf", _float_is_inf, DataType.bool())

    def fill_nan(self, value: float):
        return self._fn("fill_nan", _float_fill_nan,
                        lambda f: f[0].dtype, value)


def _float_is_nan(s: Series) -> Series:
    return Series(s.name, DataType.bool(), data=torch.isnan(s.data),
                  validity=s.validity)


def _float_is_inf(s: Series) -> Series:
    return Series(s.name, DataType.bool(), data=torch.isinf(s.data),
                  validity=s.validity)


def _float_fill_nan(s: Series, value: float) -> Series:
    return Series(s.name, s.dtype, data=torch.nan_to_num(s.data, nan=value),
                  validity=s.validity)


class UrlNamespace(_Namespace):
    def download(self, on_error: str = "raise",
                 max_connections: int = 32) -> Expression:
        from ..functions.url import url_download_series
        return self._fn("url_download", url_download_series,
                        DataType.binary(), on_error, max_connections)

    def upload(self, location: str, paths) -> Expression:
        from ..functions.url import url_upload_series
        return Expression(ScalarFn(
            "url_upload", url_upload_series,
            [self._node, _to_node(paths)], DataType.string(), (location,)))


class BinaryNamespace(_Namespace):
    def length(self) -> Expression:
        from ..kernels import strings as k
        return self._fn("binary_length", k.length_bytes, DataType.uint64())

    def concat(self, other) -> Expression:
        from ..kernels import strings as k
        return Expression(ScalarFn(
            "binary_concat", lambda a, b: k.concat_str([a, b]),
            [self._node, _to_node(other)], DataType.binary()))

    def slice(self, start: int, length: Optional[int] = None) -> Expression:
        from ..kernels import strings as k
        return self._fn("binary_slice", k.substr, DataType.binary(), start,
                        length)


class EmbeddingNamespace(_Namespace):
    def cosine_distance(self, other) -> Expression:
        from ..functions.distance import cosine_distance_series
        return Expression(ScalarFn(
            "cosine_distance", cosine_distance_series,
            [self._node, _to_node(other)], DataType.float64()))

    def dot(self, other) -> Expression:
        from ..functions.distance import dot_series
        return Expression(ScalarFn(
            "dot", dot_series, [self._node, _to_node(other)],
            DataType.float64()))

    def l2_norm(self) -> Expression:
        from ..functions.distance import l2_norm_series
        return Expression(ScalarFn("l2_norm", l2_norm_series, [self._node],
                                   DataType.float64()))


# ---------------------------------------------------------------------------
# public constructors
# ---------------------------------------------------------------------------

def col(name: str) -> Expression:
    return Expression(ColumnRef(name))


def lit(value: Any, dtype: Optional[DataType] = None) -> Expression:
    return Expression(Literal(value, dtype))


def element() -> Expression:
    """Placeholder for list.map-style element references."""
    return Expression(ColumnRef("__element__"))


def interval(days: int = 0, months: int = 0, years: int = 0,
             hours: int = 0, minutes: int = 0, seconds: int = 0) -> Expression:
    """Date interval literal — round 1 supports day-granularity offsets
    (month/year arithmetic handled at the binary-op layer for dates)."""
    total_days = days + 30 * months + 365 * years  # calendar-approx; see dt.truncate for exact
    if months or years:
        # exact calendar intervals handled via _dt_add_interval at eval
        return Expression(Literal({"days": days, "months": months,
                                   "years": years}, DataType.python()))
    us = ((hours * 60 + minutes) * 60 + seconds) * 1_000_000
    if us:
        return Expression(Literal(total_days * 86_400_000_000 + us,
                                  DataType.duration("us")))
    return Expression(Literal(total_days, DataType.int32()))


def list_(*exprs) -> Expression:
    nodes = [_to_node(e) for e in exprs]

    def make(*series: Series) -> Series:
        n = max(len(s) for s in series)
        series = [s.broadcast(n) if len(s) == 1 else s for s in series]
        from ..schema import supertype as st
        dt = series[0].dtype
        for s in series[1:]:
            dt = st(dt, s.dtype)
        series = [s.cast(dt) for s in series]
        k = len(series)
        # interleave rows: out child = row-major [n, k]
        idx = torch.arange(n * k, device=series[0].device)
        from ..series import Series as S
        from .. import kernels
        stacked = kernels.concat(series)
        # stacked is column-major (all of s0, then s1...) -> gather interleave
        gather = (idx % k) * n + torch.div(idx, k, rounding_mode="floor")
        child = stacked.take(gather)
        offs = torch.arange(0, (n + 1) * k, k, dtype=torch.int64,
                            device=series[0].device)
        return S(series[0].name, DataType.list(dt), offsets=offs,
                 children=[child.rename("item")])

    return Expression(ScalarFn(
        "list", make, nodes,
        lambda f: DataType.list(f[0].dtype)))


def struct(*exprs) -> Expression:
    nodes = [_to_node(e) for e in exprs]

    def make(*series: Series) -> Series:
        n = max(len(s) for s in series)
        series = [s.broadcast(n) if len(s) == 1 else s for s in series]
        from ..series import Series as S
        dt = DataType.struct({s.name: s.dtype for s in series})
        return S("struct", dt, children=list(series), length=n)

    def ret(fields):
        return DataType.struct({f.name: f.dtype for f in fields})

    return Expression(ScalarFn("struct", make, nodes, ret))


def resolve_exprs(exprs: Sequence[Union[Expression, str]]) -> List[ExprNode]:
    out = []
    for e in exprs:
        if isinstance(e, str):
            out.append(ColumnRef(e))
        elif isinstance(e, Expression):
            out.append(e._node)
        elif isinstance(e, ExprNode):
            out.append(e)
        else:
            out.append(Literal(e))
    return out
