"""Scalar math functions (capability of the reference's
daft/functions/numeric.py + daft-functions/src/numeric — free-function
forms over torch elementwise kernels; fused on device by torch)."""
from __future__ import annotations

import math as _m

import torch

from ..expressions.expressions import Expression, ScalarFn, _to_node
from ..schema import DataType
from ..series import Series


def _f64(name):
    def deco(fn):
        def make(expr, *args):
            def run(s: Series, *extra) -> Series:
                d = s.data.to(torch.float64)
                out = fn(d, *extra)
                return Series(s.name, DataType.float64(), data=out,
                              validity=s.validity)
            nodes = [_to_node(expr)]
            return Expression(ScalarFn(name, run, nodes,
                                       DataType.float64(), tuple(args)))
        make.__name__ = name
        return make
    return deco


def _f64_2(name):
    def deco(fn):
        def make(a, b):
            def run(x: Series, y: Series) -> Series:
                xd = x.data.to(torch.float64)
                yd = y.data.to(torch.float64)
                out = fn(xd, yd)
                v = x.validity
                if y.validity is not None:
                    v = y.validity if v is None else (v & y.validity)
                return Series(x.name, DataType.float64(), data=out,
                              validity=v)
            return Expression(ScalarFn(name, run,
                                       [_to_node(a), _to_node(b)],
                                       DataType.float64()))
        make.__name__ = name
        return make
    return deco


sin = _f64("sin")(torch.sin)
cos = _f64("cos")(torch.cos)
tan = _f64("tan")(torch.tan)
cot = _f64("cot")(lambda d: 1.0 / torch.tan(d))
sec = _f64("sec")(lambda d: 1.0 / torch.cos(d))
csc = _f64("csc")(lambda d: 1.0 / torch.sin(d))
sinh = _f64("sinh")(torch.sinh)
cosh = _f64("cosh")(torch.cosh)
tanh = _f64("tanh")(torch.tanh)
arcsin = _f64("arcsin")(torch.asin)
arccos = _f64("arccos")(torch.acos)
arctan = _f64("arctan")(torch.atan)
arcsinh = _f64("arcsinh")(torch.asinh)
arccosh = _f64("arccosh")(torch.acosh)
arctanh = _f64("arctanh")(torch.atanh)
degrees = _f64("degrees")(torch.rad2deg)
radians = _f64("radians")(torch.deg2rad)
exp = _f64("exp")(torch.exp)
expm1 = _f64("expm1")(torch.expm1)
ln = _f64("ln")(torch.log)
log2 = _f64("log2")(torch.log2)
log10 = _f64("log10")(torch.log10)
log1p = _f64("log1p")(torch.log1p)
sqrt = _f64("sqrt")(torch.sqrt)
cbrt = _f64("cbrt")(lambda d: torch.sign(d) * torch.pow(d.abs(), 1 / 3))
arctan2 = _f64_2("arctan2")(torch.atan2)
hypot = _f64_2("hypot")(torch.hypot)


def log(expr, base: float = _m.e):
    return _f64("log")(lambda d: torch.log(d) / _m.log(base))(expr)


def sign(expr):
    def run(s: Series) -> Series:
        return Series(s.name, s.dtype, data=torch.sign(s.data),
                      validity=s.validity)
    return Expression(ScalarFn("sign", run, [_to_node(expr)],
                               lambda f: f[0].dtype))


signum = sign


def negate(expr):
    from ..expressions.expressions import Expression as E
    e = expr if isinstance(expr, E) else E(_to_node(expr))
    return -e


negative = negate


def pmod(a, b):
    """Positive modulo (result sign follows the divisor, as in Spark)."""
    from ..expressions.expressions import Expression as E, BinaryOp
    return E(BinaryOp("mod", _to_node(a), _to_node(b)))


def power(a, b):
    from ..expressions.expressions import Expression as E, BinaryOp
    return E(BinaryOp("pow", _to_node(a), _to_node(b)))


pow = power  # noqa: A001


def factorial(expr):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else _m.factorial(int(v)) for v in vals]
        r = Series.from_pylist(s.name, out, DataType.int64())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("factorial", run, [_to_node(expr)],
                               DataType.int64()))


def e():
    from ..expressions.expressions import lit
    return lit(_m.e)


def pi():
    from ..expressions.expressions import lit
    return lit(_m.pi)


def trunc(expr):
    def run(s: Series) -> Series:
        return Series(s.name, DataType.float64(),
                      data=torch.trunc(s.data.to(torch.float64)),
                      validity=s.validity)
    return Expression(ScalarFn("trunc", run, [_to_node(expr)],
                               DataType.float64()))


def _bitwise(name, fn):
    def make(a, b):
        def run(x: Series, y: Series) -> Series:
            out = fn(x.data.to(torch.int64), y.data.to(torch.int64))
            v = x.validity
            if y.validity is not None:
                v = y.validity if v is None else (v & y.validity)
            return Series(x.name, DataType.int64(), data=out, validity=v)
        return Expression(ScalarFn(name, run, [_to_node(a), _to_node(b)],
                                   DataType.int64()))
    make.__name__ = name
    return make


bitwise_and = _bitwise("bitwise_and", torch.bitwise_and)
bitwise_or = _bitwise("bitwise_or", torch.bitwise_or)
bitwise_xor = _bitwise("bitwise_xor", torch.bitwise_xor)
shift_left = _bitwise("shift_left", torch.bitwise_left_shift)
shift_right = _bitwise("shift_right", torch.bitwise_right_shift)


def try_divide(a, b):
    """Division that yields null (not inf/error) on zero divisors."""
    def run(x: Series, y: Series) -> Series:
        xd = x.data.to(torch.float64)
        yd = y.data.to(torch.float64)
        ok = yd != 0
        out = xd / torch.where(ok, yd, torch.ones_like(yd))
        if ok.numel() == 1 and xd.numel() > 1:
            ok = ok.expand(xd.numel())
        v = ok.clone()
        if x.validity is not None:
            v &= x.validity
        if y.validity is not None:
            v &= y.validity if y.validity.numel() != 1 \
                else y.validity.expand(v.numel())
        return Series(x.name, DataType.float64(),
                      data=out.expand(v.numel()) if out.numel() == 1 and
                      v.numel() > 1 else out, validity=v)
    return Expression(ScalarFn("try_divide", run,
                               [_to_node(a), _to_node(b)],
                               DataType.float64()))


def random_int(expr, low: int = 0, high: int = 2**31 - 1):
    """Deterministic per-row pseudo-random int64 derived from the input's
    row hash (offline-reproducible, unlike a host RNG)."""
    from ..kernels import rowops

    def run(s: Series) -> Series:
        h = rowops.hash_columns([s]).abs()
        out = low + (h % max(1, high - low))
        return Series(s.name, DataType.int64(), data=out,
                      validity=s.validity)
    return Expression(ScalarFn("random_int", run, [_to_node(expr)],
                               DataType.int64()))
