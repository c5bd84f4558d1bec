"""Free-function aliases for namespace methods + the remaining odds and
ends of the reference's daft/functions export list."""
from __future__ import annotations

import datetime as _dt

import torch

from ..expressions.expressions import Expression, ScalarFn, _to_node, lit
from ..schema import DataType, TypeKind
from ..series import Series


def _e(x) -> Expression:
    return x if isinstance(x, Expression) else Expression(_to_node(x))


# -- str namespace delegations ---------------------------------------------
def lower(x): return _e(x).str.lower()
def upper(x): return _e(x).str.upper()
def capitalize(x): return _e(x).str.capitalize()
def strip(x): return _e(x).str.strip()
def lstrip(x): return _e(x).str.lstrip()
def rstrip(x): return _e(x).str.rstrip()
def reverse(x): return _e(x).str.reverse()
def length(x): return _e(x).str.length()
def length_bytes(x): return _e(x).str.length_bytes()
def contains(x, pat): return _e(x).str.contains(pat)
def startswith(x, pat): return _e(x).str.startswith(pat)
def endswith(x, pat): return _e(x).str.endswith(pat)
def like(x, pat): return _e(x).str.like(pat)
def ilike(x, pat): return _e(x).str.ilike(pat)
def find(x, sub): return _e(x).str.find(sub)
def split(x, sep): return _e(x).str.split(sep)
def substr(x, start, length=None): return _e(x).str.substr(start, length)
def left(x, n): return _e(x).str.left(n)
def right(x, n): return _e(x).str.right(n)
def lpad(x, n, pad=" "): return _e(x).str.lpad(n, pad)
def rpad(x, n, pad=" "): return _e(x).str.rpad(n, pad)
def repeat(x, n): return _e(x).str.repeat(n)
def concat(a, b): return _e(a).str.concat(b)
def tokenize_encode(x, tokenizer="simple"):
    return _e(x).str.tokenize_encode(tokenizer)


def tokenize_decode(x, tokenizer="simple"):
    from .tokenize import tokenize_decode_series
    return Expression(ScalarFn(
        "tokenize_decode", tokenize_decode_series, [_to_node(x)],
        DataType.string(), (tokenizer,)))


# -- float / list / struct / dt delegations ---------------------------------
def is_nan(x): return _e(x).float.is_nan()
def is_inf(x): return _e(x).float.is_inf()
def fill_nan(x, v): return _e(x).float.fill_nan(v)
def get(x, key, default=None): return _e(x).list.get(key, default)
def chunk(x, n): return _e(x).list.chunk(n)
def slice(x, a, b=None): return _e(x).list.slice(a, b)  # noqa: A001
def value_counts(x): return _e(x).list.value_counts()
def explode(x): return _Explode(_e(x))
def date(x): return _e(x).dt.date()
def total_days(x): return _e(x).dt.total_days()
def total_seconds(x): return _e(x).dt.total_seconds()


def time(x):
    """Time-of-day in microseconds since midnight."""
    from .temporal import _ts_sub_us

    def run(s: Series) -> Series:
        us = _ts_sub_us(s)
        return Series(s.name, DataType.int64(),
                      data=torch.remainder(us, 86_400_000_000),
                      validity=s.validity)
    return Expression(ScalarFn("time", run, [_to_node(x)],
                               DataType.int64()))


# -- url / image / ai delegations -------------------------------------------
def download(x, **kw): return _e(x).url.download(**kw)
def upload(x, location, name=None): return _e(x).url.upload(location, name)
def resize(x, h, w): return _e(x).image.resize(h, w)
def crop(x, *a): return _e(x).image.crop(*a)


def image_file(x):
    from . import file as _file
    return _file(x)


def image_file_metadata(x):
    """File -> {mime_type, size} struct for image files."""
    def run(s: Series) -> Series:
        mts, szs = [], []
        for f in (s.pyobjs or []):
            mts.append(None if f is None else f.mime_type())
            szs.append(None if f is None else f.size())
        ch = [Series.from_pylist("mime_type", mts, DataType.string()),
              Series.from_pylist("size", szs, DataType.int64())]
        return Series(s.name, DataType.struct(
            {"mime_type": DataType.string(), "size": DataType.int64()}),
            children=ch, length=len(s))
    return Expression(ScalarFn("image_file_metadata", run, [_to_node(x)],
                               DataType.struct(
                                   {"mime_type": DataType.string(),
                                    "size": DataType.int64()})))


def decode_image_file(x, mode: str = "RGB"):
    """File column -> decoded Image (reads each file's bytes first)."""
    def run(s: Series) -> Series:
        from .image import decode_series
        blobs = [None if f is None else f.read() for f in (s.pyobjs or [])]
        bs = Series.from_pylist(s.name, blobs, DataType.binary())
        return decode_series(bs, mode)
    return Expression(ScalarFn("decode_image_file", run, [_to_node(x)],
                               DataType.image()))


# -- distances over embeddings ----------------------------------------------
def cosine_distance(a, b): return _e(a).embedding.cosine_distance(_e(b))


def _emb_pair(name, fn, out_dt=None):
    def make(a, b):
        def run(x: Series, y: Series) -> Series:
            n = len(x)
            dx = x.children[0].data.reshape(n, -1).to(torch.float64)
            dy = y.children[0].data.reshape(len(y), -1).to(torch.float64)
            if len(y) == 1 and n > 1:
                dy = dy.expand(n, -1)
            out = fn(dx, dy)
            v = x.validity
            if y.validity is not None and y.validity.numel() == n:
                v = y.validity if v is None else (v & y.validity)
            return Series(x.name, out_dt or DataType.float64(), data=out,
                          validity=v)
        return Expression(ScalarFn(name, run, [_to_node(a), _to_node(b)],
                                   out_dt or DataType.float64()))
    make.__name__ = name
    return make


dot_product = _emb_pair("dot_product", lambda a, b: (a * b).sum(dim=1))
euclidean_distance = _emb_pair(
    "euclidean_distance", lambda a, b: torch.sqrt(((a - b) ** 2).sum(dim=1)))
cosine_similarity = _emb_pair(
    "cosine_similarity",
    lambda a, b: (a * b).sum(dim=1) /
    (a.norm(dim=1) * b.norm(dim=1)).clamp(min=1e-300))
hamming_distance = _emb_pair(
    "hamming_distance", lambda a, b: (a != b).sum(dim=1),
    DataType.int64())


# -- map helpers (first-class Map + struct-backed fallback) ------------------
def _map_value_dtype(s: Series):
    return s.dtype.inner.inner.fields[1].dtype


def map_get(x, key):
    """Map access: value for `key` per row (null when absent).  Works on
    first-class Map columns (list<struct<key,value>> physical, ref
    daft-schema Map) and struct-backed maps."""
    def ret(fields):
        dt = fields[0].dtype
        if dt.kind == TypeKind.MAP:
            return dt.inner.inner.fields[1].dtype
        if dt.kind == TypeKind.STRUCT:
            for sub in dt.fields:
                if sub.name == key:
                    return sub.dtype
        return DataType.string()

    def runner(s: Series) -> Series:
        if s.dtype.kind == TypeKind.MAP:
            vals = s.cpu().to_pylist()
            out = [None if v is None else v.get(key) for v in vals]
            r = Series.from_pylist(s.name, out, _map_value_dtype(s))
            return r.to(s.device) if s.is_gpu() else r
        if s.dtype.kind == TypeKind.STRUCT:
            for i, sub in enumerate(s.dtype.fields):
                if sub.name == key:
                    c = s.children[i].rename(s.name)
                    if s.validity is not None:
                        v = c.validity & s.validity                             if c.validity is not None else s.validity
                        c = c.with_validity(v)
                    return c
            raise KeyError(key)
        raise TypeError("map_get expects a Map or struct column")
    return Expression(ScalarFn("map_get", runner, [_to_node(_e(x))], ret))


def map_keys(x):
    def run(s: Series) -> Series:
        if s.dtype.kind == TypeKind.MAP:
            vals = s.cpu().to_pylist()
            out = [None if v is None else list(v.keys()) for v in vals]
            kd = s.dtype.inner.inner.fields[0].dtype
            r = Series.from_pylist(s.name, out, DataType.list(kd))
            return r.to(s.device) if s.is_gpu() else r
        if s.dtype.kind == TypeKind.STRUCT:
            keys = [f.name for f in s.dtype.fields]
            out = [keys] * len(s)
            r = Series.from_pylist(s.name, out,
                                   DataType.list(DataType.string()))
            return r.to(s.device) if s.is_gpu() else r
        raise TypeError("map_keys expects a Map or struct column")
    return Expression(ScalarFn("map_keys", run, [_to_node(x)],
                               DataType.list(DataType.string())))


def map_values(x):
    def run(s: Series) -> Series:
        if s.dtype.kind != TypeKind.MAP:
            raise TypeError("map_values expects a Map column")
        vals = s.cpu().to_pylist()
        out = [None if v is None else list(v.values()) for v in vals]
        vd = s.dtype.inner.inner.fields[1].dtype
        r = Series.from_pylist(s.name, out, DataType.list(vd))
        return r.to(s.device) if s.is_gpu() else r
    def ret(fields):
        dt = fields[0].dtype
        if dt.kind == TypeKind.MAP:
            return DataType.list(dt.inner.inner.fields[1].dtype)
        return DataType.list(DataType.string())
    return Expression(ScalarFn("map_values", run, [_to_node(x)], ret))


# -- timezone conversions ----------------------------------------------------
def _tz_shift(name, to_utc):
    def make(x, tz: str):
        def run(s: Series) -> Series:
            from zoneinfo import ZoneInfo
            vals = s.cpu().to_pylist()
            z = ZoneInfo(tz)
            out = []
            for v in vals:
                if v is None:
                    out.append(None)
                elif to_utc:
                    out.append(v.replace(tzinfo=z)
                               .astimezone(_dt.timezone.utc)
                               .replace(tzinfo=None))
                else:
                    out.append(v.replace(tzinfo=_dt.timezone.utc)
                               .astimezone(z).replace(tzinfo=None))
            r = Series.from_pylist(s.name, out, DataType.timestamp("us"))
            return r.to(s.device) if s.is_gpu() else r
        return Expression(ScalarFn(name, run, [_to_node(x)],
                                   DataType.timestamp("us")))
    make.__name__ = name
    return make


to_utc_timestamp = _tz_shift("to_utc_timestamp", True)
from_utc_timestamp = _tz_shift("from_utc_timestamp", False)
convert_time_zone = from_utc_timestamp
convert_timezone = from_utc_timestamp


def replace_time_zone(x, tz):
    # naive timestamps: metadata-only change
    def run(s: Series) -> Series:
        return Series(s.name, DataType.timestamp(s.dtype.timeunit or "us",
                                                 tz),
                      data=s.data, validity=s.validity)
    return Expression(ScalarFn("replace_time_zone", run, [_to_node(x)],
                               lambda f: DataType.timestamp(
                                   f[0].dtype.timeunit or "us", tz)))


def make_timestamp(y, mo, d, h, mi, s_, tz=None):
    def run(*cols) -> Series:
        import builtins
        vs = [c.cpu().data.to(torch.int64).numpy() for c in cols]
        n = builtins.max(len(v) for v in vs)
        out = []
        for i in range(n):
            g = [int(v[i % len(v)]) for v in vs]
            out.append(_dt.datetime(g[0], g[1], g[2], g[3], g[4], g[5]))
        r = Series.from_pylist(cols[0].name, out, DataType.timestamp("us"))
        return r.to(cols[0].device) if cols[0].is_gpu() else r
    return Expression(ScalarFn(
        "make_timestamp", run,
        [_to_node(v) for v in (y, mo, d, h, mi, s_)],
        DataType.timestamp("us", tz)))


make_timestamp_ltz = make_timestamp


# -- partitioning transforms (iceberg-style) ---------------------------------
def partition_days(x): return _e(x).dt.date()
def partition_months(x):
    """Months since epoch 1970-01 (iceberg transform; ref:
    Expression.partition_months)."""
    e = _e(x)
    return ((e.dt.year() - 1970) * 12 + e.dt.month() - 1).alias("months")


def partition_years(x):
    """Years since epoch 1970 (iceberg transform)."""
    return (_e(x).dt.year() - 1970).alias("years")
def partition_hours(x):
    from .temporal import to_unix_epoch
    return (to_unix_epoch(x, "s") // 3600).alias("hours")


def partition_iceberg_bucket(x, n: int):
    def run(s: Series) -> Series:
        from ..kernels import rowops
        h = rowops.hash_columns([s])
        out = torch.remainder(h.abs(), n).to(torch.int32)
        return Series(s.name, DataType.int32(), data=out,
                      validity=s.validity)
    return Expression(ScalarFn("iceberg_bucket", run, [_to_node(x)],
                               DataType.int32()))


def partition_iceberg_truncate(x, w: int):
    def run(s: Series) -> Series:
        if s.dtype.kind in (TypeKind.STRING, TypeKind.BINARY) or \
                s.is_dict():
            vals = s.cpu().to_pylist()
            out = [None if v is None else v[:w] for v in vals]
            r = Series.from_pylist(s.name, out, s.dtype)
            return r.to(s.device) if s.is_gpu() else r
        d = s.data.to(torch.int64)
        out = d - torch.remainder(d, w)
        return Series(s.name, DataType.int64(), data=out,
                      validity=s.validity)
    return Expression(ScalarFn("iceberg_truncate", run, [_to_node(x)],
                               lambda f: f[0].dtype))


# -- misc ---------------------------------------------------------------------
def seq(start: int, end: int, step: int = 1):
    """Sequence literal as a list expression."""
    return lit(list(range(start, end, step)))


def bin(x):  # noqa: A001
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else format(int(v), "b") for v in vals]
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("bin", run, [_to_node(x)],
                               DataType.string()))


def conv(x, from_base: int, to_base: int):
    digits = "0123456789abcdefghijklmnopqrstuvwxyz"

    def enc(n: int, base: int) -> str:
        import builtins
        if n == 0:
            return "0"
        neg = n < 0
        n = builtins.abs(n)
        out = ""
        while n:
            out = digits[n % base] + out
            n //= base
        return ("-" if neg else "") + out

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else enc(int(str(v), from_base), to_base)
               for v in vals]
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("conv", run, [_to_node(x)],
                               DataType.string()))


class _Unnest:
    """Marker consumed by DataFrame.select: expands a struct column into
    one output column per field (ref: daft.functions.unnest)."""

    def __init__(self, expr):
        self.expr = expr


def unnest(x):
    return _Unnest(_e(x))


class _Explode:
    """Marker consumed by DataFrame.select: the column's lists expand to
    one row per element, other columns repeating (ref:
    daft.functions.explode / Expression.explode)."""

    def __init__(self, expr, name=None):
        self.expr = expr
        self.name = name

    def alias(self, name):
        return _Explode(self.expr, name)


def first_value(x, ignore_nulls: bool = False):
    """first_value WINDOW function (use with .over(window); ref:
    daft/functions/window.py:310)."""
    from . import w_first_value
    return w_first_value(_e(x))


def last_value(x, ignore_nulls: bool = False):
    from . import w_last_value
    return w_last_value(_e(x))


def jq(x, filter_expr: str):
    from .misc import jq as _jq
    return _jq(_e(x), filter_expr)


def _uuid7_part(name, fn):
    def make(x):
        def run(s: Series) -> Series:
            vals = s.cpu().to_pylist()
            out = []
            for v in vals:
                if v is None:
                    out.append(None)
                    continue
                hx = v.replace("-", "")
                ms = int(hx[:12], 16)
                ts = _dt.datetime.utcfromtimestamp(ms / 1000.0)
                out.append(fn(ts))
            r = Series.from_pylist(s.name, out, DataType.int32())
            return r.to(s.device) if s.is_gpu() else r
        return Expression(ScalarFn(name, run, [_to_node(x)],
                                   DataType.int32()))
    make.__name__ = name
    return make


extract_month_uuid7 = _uuid7_part("extract_month_uuid7", lambda t: t.month)
extract_day_uuid7 = _uuid7_part("extract_day_uuid7", lambda t: t.day)
extract_hour_uuid7 = _uuid7_part("extract_hour_uuid7", lambda t: t.hour)
extract_minute_uuid7 = _uuid7_part("extract_minute_uuid7",
                                   lambda t: t.minute)


def resample(x, every: str):
    """Truncate timestamps to a resampling interval (group key helper)."""
    return _e(x).dt.truncate(every)


# -- free-function forms of Expression methods (ref: daft/functions
# exports the method surface as functions too) -------------------------------

def abs(x): return _e(x).abs()                     # noqa: A001
def any_value(x): return _e(x).any_value()
def approx_count_distinct(x): return _e(x).approx_count_distinct()
def avg(x): return _e(x).avg()
def between(x, lo, hi): return _e(x).between(lo, hi)
def bool_and(x): return _e(x).bool_and()
def bool_or(x): return _e(x).bool_or()
def cast(x, dtype): return _e(x).cast(dtype)
def ceil(x): return _e(x).ceil()
def clip(x, lo=None, hi=None): return _e(x).clip(lo, hi)
def count(x=None):
    """count(expr) counts non-null rows of expr; count() is COUNT(*)."""
    if x is None:
        from ..expressions.expressions import Agg, AggKind, Expression
        return Expression(Agg(AggKind.COUNT_ALL, None))
    return _e(x).count()
def count_distinct(x): return _e(x).count_distinct()
def fill_null(x, v): return _e(x).fill_null(v)
def floor(x): return _e(x).floor()
def hash(x, seed: int = 0): return _e(x).hash(seed)        # noqa: A001
def is_in(x, values): return _e(x).is_in(values)
def is_null(x): return _e(x).is_null()
def lag(x, n: int = 1, default=None): return _e(x).lag(n, default)
def lead(x, n: int = 1, default=None): return _e(x).lead(n, default)
def max(x): return _e(x).max()                     # noqa: A001
def mean(x): return _e(x).mean()
def min(x): return _e(x).min()                     # noqa: A001
def minhash(x, num_hashes: int, ngram_size: int = 1, seed: int = 1):
    return _e(x).minhash(num_hashes, ngram_size, seed)
def not_null(x): return _e(x).not_null()
def over(x, window): return _e(x).over(window)
def round(x, decimals: int = 0): return _e(x).round(decimals)  # noqa: A001
def simhash(x, ngram_size: int = 4): return _e(x).simhash(ngram_size)
def skew(x): return _e(x).skew()
def stddev(x): return _e(x).stddev()
def sum(x): return _e(x).sum()                     # noqa: A001
