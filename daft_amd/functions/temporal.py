"""Temporal free functions (capability of daft/functions/temporal.py +
daft-functions-temporal).  Most delegate to the dt namespace; the rest
are vectorized here."""
from __future__ import annotations

import datetime as _dt

import torch

from ..expressions.expressions import Expression, ScalarFn, _to_node, lit
from ..schema import DataType, TypeKind
from ..series import Series


def _e(x) -> Expression:
    return x if isinstance(x, Expression) else Expression(_to_node(x))


def year(x): return _e(x).dt.year()
def month(x): return _e(x).dt.month()
def day(x): return _e(x).dt.day()
def dayofmonth(x): return _e(x).dt.day()
def day_of_month(x): return _e(x).dt.day()
def dayofyear(x): return _e(x).dt.day_of_year()
def day_of_year(x): return _e(x).dt.day_of_year()
def weekofyear(x): return _e(x).dt.week_of_year()
def week_of_year(x): return _e(x).dt.week_of_year()
def quarter(x): return _e(x).dt.quarter()
def hour(x): return _e(x).dt.hour()
def minute(x): return _e(x).dt.minute()
def second(x): return _e(x).dt.second()
def day_of_week(x): return _e(x).dt.day_of_week()
def dayofweek(x): return _e(x).dt.day_of_week()
def date_trunc(interval: str, x): return _e(x).dt.truncate(interval)
def to_date(x, fmt: str = "%Y-%m-%d"): return _e(x).str.to_date(fmt)


def _ts_sub_us(s: Series) -> torch.Tensor:
    unit = s.dtype.timeunit or "us"
    mul = {"s": 1_000_000, "ms": 1_000, "us": 1, "ns": 1}[unit]
    v = s.data.to(torch.int64)
    if unit == "ns":
        v = torch.div(v, 1000, rounding_mode="floor")
    else:
        v = v * mul
    return v


def _sub_second(name, modulus, divisor):
    def make(x):
        def run(s: Series) -> Series:
            us = _ts_sub_us(s)
            out = torch.remainder(us, 1_000_000) // divisor
            return Series(s.name, DataType.int32(),
                          data=out.to(torch.int32), validity=s.validity)
        return Expression(ScalarFn(name, run, [_to_node(x)],
                                   DataType.int32()))
    make.__name__ = name
    return make


microsecond = _sub_second("microsecond", 1_000_000, 1)
millisecond = _sub_second("millisecond", 1_000_000, 1000)


def nanosecond(x):
    def run(s: Series) -> Series:
        unit = s.dtype.timeunit or "us"
        v = s.data.to(torch.int64)
        mul = {"s": 10**9, "ms": 10**6, "us": 10**3, "ns": 1}[unit]
        out = torch.remainder(v * mul, 10**9)
        return Series(s.name, DataType.int64(), data=out,
                      validity=s.validity)
    return Expression(ScalarFn("nanosecond", run, [_to_node(x)],
                               DataType.int64()))


def date_add(x, days): return _e(x) + days
def dateadd(unit, n, x):
    if unit in ("day", "days"):
        return _e(x) + n
    if unit in ("month", "months"):
        return add_months(x, n)
    if unit in ("year", "years"):
        return add_months(x, 12 * n)
    raise ValueError(f"dateadd unit {unit!r}")


def date_sub(x, days): return _e(x) - days


def date_diff(end, start, start2=None):
    """Whole days between two dates; the SQL 3-arg form
    DATEDIFF(unit, start, end) is also accepted (unit day/month/year)."""
    if start2 is not None:
        unit, a, b = end, start2, start     # DATEDIFF('d', start, end)
        if isinstance(unit, Expression):    # SQL binds literals as exprs
            from ..expressions.expressions import Literal as _L
            node = unit._node
            if isinstance(node, _L):
                unit = node.value
        unit = str(unit).rstrip("s").lower()
        if unit in ("day", "d"):
            return date_diff(a, b)
        if unit == "month":
            ey, em = _e(a).dt.year(), _e(a).dt.month()
            sy, sm = _e(b).dt.year(), _e(b).dt.month()
            return (ey - sy) * 12 + (em - sm)
        if unit == "year":
            return _e(a).dt.year() - _e(b).dt.year()
        raise ValueError(f"datediff unit {unit!r}")

    def run(a: Series, b: Series) -> Series:
        out = a.data.to(torch.int64) - b.data.to(torch.int64)
        v = a.validity
        if b.validity is not None:
            v = b.validity if v is None else (v & b.validity)
        return Series(a.name, DataType.int64(), data=out, validity=v)
    return Expression(ScalarFn("date_diff", run,
                               [_to_node(end), _to_node(start)],
                               DataType.int64()))


datediff = date_diff


def datepart(part: str, x):
    fns = {"year": year, "month": month, "day": day, "hour": hour,
           "minute": minute, "second": second, "quarter": quarter,
           "week": weekofyear, "dow": day_of_week, "doy": dayofyear}
    if part.lower() not in fns:
        raise ValueError(f"datepart {part!r}")
    return fns[part.lower()](x)


def make_date(y, m, d):
    def run(ys: Series, ms: Series, ds: Series) -> Series:
        import numpy as np
        yv = ys.cpu().data.to(torch.int64).numpy()
        mv = ms.cpu().data.to(torch.int64).numpy()
        dv = ds.cpu().data.to(torch.int64).numpy()
        n = max(len(yv), len(mv), len(dv))
        out = []
        for i in range(n):
            out.append(_dt.date(int(yv[i % len(yv)]), int(mv[i % len(mv)]),
                                int(dv[i % len(dv)])))
        r = Series.from_pylist(ys.name, out, DataType.date())
        return r.to(ys.device) if ys.is_gpu() else r
    return Expression(ScalarFn("make_date", run,
                               [_to_node(y), _to_node(m), _to_node(d)],
                               DataType.date()))


def last_day(x):
    """Last day of the month of each date."""
    def run(s: Series) -> Series:
        import calendar
        vals = s.cpu().to_pylist()
        out = [None if v is None else
               v.replace(day=calendar.monthrange(v.year, v.month)[1])
               for v in vals]
        r = Series.from_pylist(s.name, out, DataType.date())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("last_day", run, [_to_node(x)],
                               DataType.date()))


def next_day(x, dow: str):
    names = ["monday", "tuesday", "wednesday", "thursday", "friday",
             "saturday", "sunday"]
    target = names.index(dow.lower())

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
                continue
            delta = (target - v.weekday()) % 7 or 7
            out.append(v + _dt.timedelta(days=delta))
        r = Series.from_pylist(s.name, out, DataType.date())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("next_day", run, [_to_node(x)],
                               DataType.date()))


def add_months(x, n: int):
    def run(s: Series) -> Series:
        import calendar
        vals = s.cpu().to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
                continue
            mo = v.month - 1 + n
            y = v.year + mo // 12
            m = mo % 12 + 1
            out.append(_dt.date(y, m, min(v.day,
                                          calendar.monthrange(y, m)[1])))
        r = Series.from_pylist(s.name, out, DataType.date())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("add_months", run, [_to_node(x)],
                               DataType.date()))


def months_between(end, start):
    def run(a: Series, b: Series) -> Series:
        av = a.cpu().to_pylist()
        bv = b.cpu().to_pylist()
        out = []
        for x, y in zip(av, bv):
            if x is None or y is None:
                out.append(None)
            else:
                out.append((x.year - y.year) * 12 + (x.month - y.month)
                           + (x.day - y.day) / 31.0)
        r = Series.from_pylist(a.name, out, DataType.float64())
        return r.to(a.device) if a.is_gpu() else r
    return Expression(ScalarFn("months_between", run,
                               [_to_node(end), _to_node(start)],
                               DataType.float64()))


def strftime(x, fmt: str = "%Y-%m-%d"):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else v.strftime(fmt) for v in vals]
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("strftime", run, [_to_node(x)],
                               DataType.string()))


date_format = strftime


def from_unixtime(x, unit: str = "s"):
    def run(s: Series) -> Series:
        mul = {"s": 10**6, "ms": 10**3, "us": 1}[unit]
        out = s.data.to(torch.int64) * mul
        return Series(s.name, DataType.timestamp("us"), data=out,
                      validity=s.validity)
    return Expression(ScalarFn("from_unixtime", run, [_to_node(x)],
                               DataType.timestamp("us")))


def timestamp_seconds(x): return from_unixtime(x, "s")
def timestamp_millis(x): return from_unixtime(x, "ms")
def timestamp_micros(x): return from_unixtime(x, "us")


def to_unix_epoch(x, unit: str = "s"):
    def run(s: Series) -> Series:
        if s.dtype.kind == TypeKind.DATE:
            us = s.data.to(torch.int64) * 86_400_000_000
        else:
            us = _ts_sub_us(s)
        div = {"s": 10**6, "ms": 10**3, "us": 1}[unit]
        return Series(s.name, DataType.int64(),
                      data=torch.div(us, div, rounding_mode="floor"),
                      validity=s.validity)
    return Expression(ScalarFn("to_unix_epoch", run, [_to_node(x)],
                               DataType.int64()))


def unix_date(x):
    def run(s: Series) -> Series:
        if s.dtype.kind == TypeKind.DATE:
            d = s.data.to(torch.int64)
        else:
            d = torch.div(_ts_sub_us(s), 86_400_000_000,
                          rounding_mode="floor")
        return Series(s.name, DataType.int64(), data=d,
                      validity=s.validity)
    return Expression(ScalarFn("unix_date", run, [_to_node(x)],
                               DataType.int64()))


def date_from_unix_date(x):
    def run(s: Series) -> Series:
        return Series(s.name, DataType.date(),
                      data=s.data.to(torch.int32), validity=s.validity)
    return Expression(ScalarFn("date_from_unix_date", run, [_to_node(x)],
                               DataType.date()))


def current_date():
    return lit(_dt.date.today())


def current_timestamp():
    return lit(_dt.datetime.now())


def current_timezone():
    import time
    return lit(time.tzname[0])


def _total(name, div):
    def make(x):
        def run(s: Series) -> Series:
            # duration stored in us
            out = torch.div(s.data.to(torch.int64), div,
                            rounding_mode="floor")
            return Series(s.name, DataType.int64(), data=out,
                          validity=s.validity)
        return Expression(ScalarFn(name, run, [_to_node(x)],
                                   DataType.int64()))
    make.__name__ = name
    return make


total_hours = _total("total_hours", 3_600_000_000)
total_minutes = _total("total_minutes", 60_000_000)
total_milliseconds = _total("total_milliseconds", 1_000)
total_microseconds = _total("total_microseconds", 1)


def total_nanoseconds(x):
    def run(s: Series) -> Series:
        return Series(s.name, DataType.int64(),
                      data=s.data.to(torch.int64) * 1000,
                      validity=s.validity)
    return Expression(ScalarFn("total_nanoseconds", run, [_to_node(x)],
                               DataType.int64()))


def to_datetime(x, fmt: str = "%Y-%m-%d %H:%M:%S"):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else _dt.datetime.strptime(v, fmt)
               for v in vals]
        r = Series.from_pylist(s.name, out, DataType.timestamp("us"))
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("to_datetime", run, [_to_node(x)],
                               DataType.timestamp("us")))
