"""Misc free functions: list aliases/extras, serde/codec, json helpers,
when() case builder, aggregate helpers (capability of the corresponding
daft/functions modules; see each docstring)."""
from __future__ import annotations

import json as _json
from typing import Optional

import torch

from ..expressions.expressions import (Agg, AggKind, Expression, ScalarFn,
                                       _to_node, lit)
from ..schema import DataType, TypeKind
from ..series import Series


def _e(x) -> Expression:
    return x if isinstance(x, Expression) else Expression(_to_node(x))


# -- list aliases / extras --------------------------------------------------

def list_contains(x, v): return _e(x).list.contains(v)
def list_distinct(x): return _e(x).list.distinct()
def list_join(x, sep): return _e(x).list.join(sep)
def list_sum(x): return _e(x).list.sum()
def list_min(x): return _e(x).list.min()
def list_max(x): return _e(x).list.max()
def list_mean(x): return _e(x).list.mean()
def list_count(x, mode: str = "valid"):
    """Count list elements: mode 'valid' (default, non-null), 'all', or
    'null' (ref: daft list_count CountMode)."""
    if mode == "all":
        return _e(x).list.length()
    def run(s):
        from ..series import Series
        from ..schema import DataType
        vals = s.cpu().to_pylist()
        if mode == "valid":
            out = [None if v is None else sum(e is not None for e in v)
                   for v in vals]
        else:
            out = [None if v is None else sum(e is None for e in v)
                   for v in vals]
        r = Series.from_pylist(s.name, out, DataType.uint64())
        return r.to(s.device) if s.is_gpu() else r
    from .aliases import _to_node
    from ..expressions.expressions import Expression, ScalarFn
    from ..schema import DataType
    return Expression(ScalarFn("list_count", run, [_to_node(_e(x))],
                               DataType.uint64()))
def list_chunk(x, n): return _e(x).list.chunk(n)
def list_slice(x, a, b=None): return _e(x).list.slice(a, b)
def list_agg(x): return _e(x).agg_list()
def to_list(x): return _e(x).agg_list()


def list_agg_distinct(x):
    return _e(x).agg_list().list.distinct()


def _list_host(name, fn, ret_dtype):
    """Per-row host mapping over a list column with a fixed result
    dtype."""
    def make(x, *args):
        def run(s: Series, *extra) -> Series:
            vals = s.cpu().to_pylist()
            out = [None if v is None else fn(v, *extra) for v in vals]
            r = Series.from_pylist(s.name, out, ret_dtype)
            return r.to(s.device) if s.is_gpu() else r
        return Expression(ScalarFn(name, run, [_to_node(x)], ret_dtype,
                                   tuple(args)))
    make.__name__ = name
    return make


def list_sort(x, desc: bool = False):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else
               sorted((u for u in v if u is not None), reverse=desc) +
               [u for u in v if u is None] for v in vals]
        r = Series.from_pylist(s.name, out, s.dtype)
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("list_sort", run, [_to_node(x)],
                               lambda f: f[0].dtype))


def list_append(x, v):
    def run(s: Series, item: Series) -> Series:
        vals = s.cpu().to_pylist()
        iv = item.cpu().to_pylist()
        out = [(a or []) + [iv[i % len(iv)]] for i, a in enumerate(vals)]
        r = Series.from_pylist(s.name, out, s.dtype)
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("list_append", run,
                               [_to_node(x), _to_node(v)],
                               lambda f: f[0].dtype))


def list_flatten(x):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else
               [u for sub in v if sub is not None for u in sub]
               for v in vals]
        inner = s.dtype.inner.inner if s.dtype.inner and \
            s.dtype.inner.kind == TypeKind.LIST else DataType.int64()
        r = Series.from_pylist(s.name, out, DataType.list(inner))
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn(
        "list_flatten", run, [_to_node(x)],
        lambda f: DataType.list(f[0].dtype.inner.inner
                                if f[0].dtype.inner is not None and
                                f[0].dtype.inner.kind == TypeKind.LIST
                                else DataType.int64())))


def list_bool_and(x):
    return _list_host("list_bool_and",
                      lambda v: (all(b for b in v if b is not None)
                                 if any(b is not None for b in v) else None),
                      DataType.bool())(x)


def list_bool_or(x):
    return _list_host("list_bool_or",
                      lambda v: (any(b for b in v if b is not None)
                                 if any(b is not None for b in v) else None),
                      DataType.bool())(x)


def list_map(x, fn):
    """Apply a python callable per element (host; for vectorized work use
    explode + expressions)."""
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else [fn(u) for u in v] for v in vals]
        r = Series.from_pylist(s.name, out)
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("list_map", run, [_to_node(x)],
                               lambda f: f[0].dtype))


def list_filter(x, fn):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else [u for u in v if fn(u)]
               for v in vals]
        r = Series.from_pylist(s.name, out, s.dtype)
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("list_filter", run, [_to_node(x)],
                               lambda f: f[0].dtype))


# -- struct / misc ----------------------------------------------------------

def to_struct(*exprs):
    """Pack columns into a struct column (ref: daft-functions to_struct)."""
    def run(*series) -> Series:
        from ..schema import Field
        names = [s.name for s in series]
        dt = DataType.struct({s.name: s.dtype for s in series})
        return Series(series[0].name, dt, children=list(series),
                      length=len(series[0]))
    return Expression(ScalarFn(
        "struct", run, [_to_node(e) for e in exprs],
        lambda f: DataType.struct({x.name: x.dtype for x in f})))


struct = to_struct


def eq_null_safe(a, b):
    """Null-safe equality: NULL <=> NULL is true (SQL IS NOT DISTINCT
    FROM)."""
    ea, eb = _e(a), _e(b)
    both_null = ea.is_null() & eb.is_null()
    neither = ea.not_null() & eb.not_null()
    return both_null | (neither & (ea == eb).fill_null(False))


def not_nan(x):
    return ~_e(x).float.is_nan()


def try_cast(x, dtype):
    """Cast that yields null instead of raising (string->numeric etc.)."""
    def run(s: Series) -> Series:
        from .. import kernels
        try:
            return kernels.cast(s, dtype)
        except Exception:
            pass
        # per-value salvage: keep the values that cast cleanly
        out = []
        for v in s.cpu().to_pylist():
            if v is None:
                out.append(None)
                continue
            try:
                one = kernels.cast(
                    Series.from_pylist("t", [v], s.dtype.cpu_like()
                                       if hasattr(s.dtype, "cpu_like")
                                       else s.dtype), dtype)
                pv = one.to_pylist()[0]
                out.append(pv)
            except Exception:
                out.append(None)
        try:
            r = Series.from_pylist(s.name, out, dtype)
        except Exception:
            from ..series import full_null
            return full_null(s.name, dtype, len(s), s.device)
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("try_cast", run, [_to_node(x)], dtype))


class _WhenThen:
    """daft.functions.when(cond, value).when(...).otherwise(default)."""

    def __init__(self, pairs):
        self._pairs = pairs

    def when(self, cond, value) -> "_WhenThen":
        return _WhenThen(self._pairs + [(cond, value)])

    def otherwise(self, default) -> Expression:
        out = default if isinstance(default, Expression) else lit(default)
        for cond, val in reversed(self._pairs):
            v = val if isinstance(val, Expression) else lit(val)
            out = _e(cond).if_else(v, out)
        return out

    # allow use without otherwise(): nulls for unmatched
    def _expr(self) -> Expression:
        return self.otherwise(lit(None))

    @property
    def _node(self):
        return self._expr()._node


def when(cond, value) -> _WhenThen:
    return _WhenThen([(cond, value)])


# -- serde / codecs ---------------------------------------------------------

def serialize(x, format: str = "json"):
    if format != "json":
        raise ValueError("serialize supports json")

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else _json.dumps(v, default=str)
               for v in vals]
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("serialize", run, [_to_node(x)],
                               DataType.string()))


def deserialize(x, format: str = "json", _safe: bool = False):
    if format != "json":
        raise ValueError("deserialize supports json")

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
                continue
            try:
                out.append(_json.loads(v))
            except Exception:
                if _safe:
                    out.append(None)
                else:
                    raise
        r = Series.from_pylist(s.name, out)
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("deserialize", run, [_to_node(x)],
                               DataType.python()))


def try_deserialize(x, format: str = "json"):
    return deserialize(x, format, _safe=True)


_CODECS = {"gzip", "gz", "zlib", "bz2"}


def _codec_fn(codec: str, do: str):
    import bz2
    import gzip
    import zlib
    c = codec.lower()
    if c in ("gzip", "gz"):
        return gzip.compress if do == "c" else gzip.decompress
    if c == "zlib":
        return zlib.compress if do == "c" else zlib.decompress
    if c == "bz2":
        return bz2.compress if do == "c" else bz2.decompress
    raise ValueError(f"unsupported codec {codec!r} (offline build: "
                     f"{sorted(_CODECS)})")


def _codec(name, do, safe=False):
    def make(x, codec: str = "zlib"):
        fn = _codec_fn(codec, do)

        def run(s: Series) -> Series:
            vals = s.cpu().to_pylist()
            out = []
            for v in vals:
                if v is None:
                    out.append(None)
                    continue
                try:
                    b = v.encode() if isinstance(v, str) else bytes(v)
                    out.append(fn(b))
                except Exception:
                    if safe:
                        out.append(None)
                    else:
                        raise
            r = Series.from_pylist(s.name, out, DataType.binary())
            return r.to(s.device) if s.is_gpu() else r
        return Expression(ScalarFn(name, run, [_to_node(x)],
                                   DataType.binary()))
    make.__name__ = name
    return make


compress = _codec("compress", "c")
decompress = _codec("decompress", "d")
try_compress = _codec("try_compress", "c", safe=True)
try_decompress = _codec("try_decompress", "d", safe=True)


def _b_codec(name, enc, safe=False):
    import base64

    def make(x, codec: str = "base64"):
        def one(v):
            b = v.encode() if isinstance(v, str) else bytes(v)
            c = codec.lower()
            if c == "base64":
                return base64.b64encode(b) if enc else base64.b64decode(b)
            if c == "hex":
                return b.hex().encode() if enc else bytes.fromhex(
                    b.decode())
            if c in ("utf-8", "utf8"):
                return b
            raise ValueError(f"unsupported encoding {codec!r}")

        def run(s: Series) -> Series:
            vals = s.cpu().to_pylist()
            out = []
            for v in vals:
                if v is None:
                    out.append(None)
                    continue
                try:
                    out.append(one(v))
                except Exception:
                    if safe:
                        out.append(None)
                    else:
                        raise
            r = Series.from_pylist(s.name, out, DataType.binary())
            return r.to(s.device) if s.is_gpu() else r
        return Expression(ScalarFn(name, run, [_to_node(x)],
                                   DataType.binary()))
    make.__name__ = name
    return make


encode = _b_codec("encode", True)
decode = _b_codec("decode", False)
try_encode = _b_codec("try_encode", True, safe=True)
try_decode = _b_codec("try_decode", False, safe=True)


# -- json helpers -----------------------------------------------------------

def json_array_length(x):
    def one(v):
        d = _json.loads(v)
        return len(d) if isinstance(d, list) else None

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else one(v) for v in vals]
        r = Series.from_pylist(s.name, out, DataType.int64())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("json_array_length", run, [_to_node(x)],
                               DataType.int64()))


def json_object_keys(x):
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = []
        for v in vals:
            if v is None:
                out.append(None)
                continue
            d = _json.loads(v)
            out.append(list(d.keys()) if isinstance(d, dict) else None)
        r = Series.from_pylist(s.name, out,
                               DataType.list(DataType.string()))
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("json_object_keys", run, [_to_node(x)],
                               DataType.list(DataType.string())))


def json_tuple(x, *fields):
    """Extract several top-level fields as string columns packed in a
    struct."""
    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        cols = {f: [] for f in fields}
        for v in vals:
            d = _json.loads(v) if v is not None else {}
            for f in fields:
                u = d.get(f) if isinstance(d, dict) else None
                cols[f].append(None if u is None else
                               (u if isinstance(u, str) else
                                _json.dumps(u)))
        children = [Series.from_pylist(f, cols[f], DataType.string())
                    for f in fields]
        dt = DataType.struct({f: DataType.string() for f in fields})
        r = Series(s.name, dt, children=children, length=len(vals))
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn(
        "json_tuple", run, [_to_node(x)],
        DataType.struct({f: DataType.string() for f in fields})))


def parse_url(x, part: Optional[str] = None):
    """Split URLs into components (scheme/host/path/query/fragment)."""
    from urllib.parse import urlparse

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        comps = {"scheme": [], "host": [], "path": [], "query": [],
                 "fragment": []}
        for v in vals:
            p = urlparse(v) if v is not None else None
            comps["scheme"].append(p.scheme if p else None)
            comps["host"].append(p.netloc if p else None)
            comps["path"].append(p.path if p else None)
            comps["query"].append(p.query if p else None)
            comps["fragment"].append(p.fragment if p else None)
        if part is not None:
            r = Series.from_pylist(s.name, comps[part], DataType.string())
        else:
            children = [Series.from_pylist(k, v, DataType.string())
                        for k, v in comps.items()]
            r = Series(s.name, DataType.struct(
                {k: DataType.string() for k in comps}), children=children,
                length=len(vals))
        return r.to(s.device) if s.is_gpu() else r
    ret = DataType.string() if part is not None else DataType.struct(
        {k: DataType.string() for k in ("scheme", "host", "path", "query",
                                        "fragment")})
    return Expression(ScalarFn("parse_url", run, [_to_node(x)], ret))


def guess_mime_type(x):
    from ..file import File

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else File(v).mime_type() for v in vals]
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("guess_mime_type", run, [_to_node(x)],
                               DataType.string()))


def file_exists(x):
    from ..file import File

    def run(s: Series) -> Series:
        out = [None if f is None else f.exists() for f in s.pyobjs]
        return Series.from_pylist(s.name, out, DataType.bool())
    return Expression(ScalarFn("file_exists", run, [_to_node(x)],
                               DataType.bool()))


def file_path(x):
    from ..file import File

    def run(s: Series) -> Series:
        out = []
        for f in s.pyobjs:
            try:
                out.append(None if f is None else f.path)
            except ValueError:
                out.append(None)
        return Series.from_pylist(s.name, out, DataType.string())
    return Expression(ScalarFn("file_path", run, [_to_node(x)],
                               DataType.string()))


# -- aggregates -------------------------------------------------------------

def median(x):
    return _e(x).approx_percentile(0.5)


def percentile(x, q: float):
    return _e(x).approx_percentile(q)


def approx_percentiles(x, qs):
    return [_e(x).approx_percentile(q).alias(f"p{q}") for q in qs]


def var(x):
    return Expression(Agg(AggKind.VARIANCE, _to_node(x)))


def product(x):
    """Product aggregate via exp(sum(log|x|)) with sign tracking is
    lossy; use an exact host fold instead (ref: daft product agg)."""
    from ..udf import udaf as _udaf

    @_udaf(return_dtype=DataType.float64())
    class _Prod:
        def aggregate(self, values):
            p = 1.0
            for v in values.to_pylist():
                if v is not None:
                    p *= v
            return p

        def combine(self, states):
            p = 1.0
            for s in states:
                p *= s
            return p

        def finalize(self, state):
            return state
    return _Prod()(x)


def string_agg(x, sep: str = ","):
    return _e(x).agg_list().list.join(sep)


def columns_mean(*exprs):
    """Row-wise mean across columns, skipping nulls."""
    es = [_e(x) for x in exprs]
    total = None
    count = None
    for e in es:
        v = e.fill_null(0.0)
        c = e.not_null().cast(DataType.int64())
        total = v if total is None else total + v
        count = c if count is None else count + c
    return total / count


def columns_sum(*exprs):
    es = [_e(x) for x in exprs]
    total = None
    for e in es:
        v = e.fill_null(0)
        total = v if total is None else total + v
    return total


def pearson_correlation(x, y):
    """Pearson r as a composed aggregate expression (sums of products)."""
    ex, ey = _e(x), _e(y)
    n = ex.count().cast(DataType.float64())
    sx = ex.sum().cast(DataType.float64())
    sy = ey.sum().cast(DataType.float64())
    sxx = (ex * ex).sum().cast(DataType.float64())
    syy = (ey * ey).sum().cast(DataType.float64())
    sxy = (ex * ey).sum().cast(DataType.float64())
    num = n * sxy - sx * sy
    import math as _mm
    den = ((n * sxx - sx * sx) * (n * syy - sy * sy))
    from .math import sqrt as _sqrt
    return num / _sqrt(den)


# -- media stubs (offline image: implemented; audio/video/hdf5: gated) ------

def _gated(name, needs):
    def make(*a, **k):
        raise RuntimeError(
            f"{name}() requires {needs}, which is not available in this "
            f"offline build")
    make.__name__ = name
    return make


audio_file = _gated("audio_file", "an audio decoding backend (soundfile)")
audio_metadata = _gated("audio_metadata", "an audio backend")
video_file = _gated("video_file", "a video backend (ffmpeg)")
video_frames = _gated("video_frames", "a video backend (ffmpeg)")
video_keyframes = _gated("video_keyframes", "a video backend (ffmpeg)")
video_metadata = _gated("video_metadata", "a video backend (ffmpeg)")
hdf5_file = _gated("hdf5_file", "h5py")
hdf5_keys = _gated("hdf5_keys", "h5py")
hdf5_attrs = _gated("hdf5_attrs", "h5py")
hdf5_metadata = _gated("hdf5_metadata", "h5py")
run_process = _gated("run_process", "subprocess execution policy")
llm_generate = _gated("llm_generate", "a local LLM provider (vllm)")


# ---------------------------------------------------------------------------
# jq-style JSON queries (ref: daft-functions-json — jaq-backed `jq` filter
# over JSON strings).  Supported subset: identity `.`, field access
# `.a.b`, optional access `.a?`, array index `.a[0]` (negative ok),
# array iteration `.a[]`, pipes `f | g`, and `//` defaults.
# ---------------------------------------------------------------------------

def _jq_compile(filter_expr: str):
    import re as _re

    def parse_one(src: str):
        steps = []
        i = 0
        src = src.strip()
        if src == ".":
            return steps
        while i < len(src):
            c = src[i]
            if c == ".":
                m = _re.match(r"\.([A-Za-z_][A-Za-z0-9_]*)(\??)",
                              src[i:])
                if m:
                    steps.append(("field", m.group(1),
                                  m.group(2) == "?"))
                    i += m.end()
                    continue
                if src[i:i + 1] == "." and src[i + 1:i + 2] == "[":
                    i += 1
                    continue
                i += 1
                continue
            if c == "[":
                m = _re.match(r"\[(-?\d*)\]", src[i:])
                if not m:
                    raise ValueError(f"jq: bad bracket at {src[i:]}")
                if m.group(1) == "":
                    steps.append(("iterate",))
                else:
                    steps.append(("index", int(m.group(1))))
                i += m.end()
                continue
            raise ValueError(f"jq filter not supported: {src!r}")
        return steps

    alts = [a.strip() for a in filter_expr.split("//")]
    progs = []
    for alt in alts:
        progs.append([parse_one(p) for p in alt.split("|")])
    return progs

def _jq_eval(steps_pipeline, doc):
    vals = [doc]
    for steps in steps_pipeline:
        for step in steps:
            out = []
            for v in vals:
                if step[0] == "field":
                    if isinstance(v, dict):
                        out.append(v.get(step[1]))
                    elif not step[2]:
                        raise TypeError("jq: cannot index non-object")
                elif step[0] == "index":
                    if isinstance(v, list):
                        try:
                            out.append(v[step[1]])
                        except IndexError:
                            out.append(None)
                    else:
                        out.append(None)
                else:  # iterate
                    if isinstance(v, list):
                        out.extend(v)
                    elif isinstance(v, dict):
                        out.extend(v.values())
            vals = out
    return vals


def jq(x, filter_expr: str):
    """jq-style filter over a JSON string column; multi-value results
    (array iteration) come back as a JSON array string (ref:
    daft-functions-json/src — jaq filters)."""
    progs = _jq_compile(filter_expr)

    def run(s: Series) -> Series:
        out = []
        for v in s.cpu().to_pylist():
            if v is None:
                out.append(None)
                continue
            try:
                doc = _json.loads(v)
            except Exception:
                out.append(None)
                continue
            res = None
            for prog in progs:
                try:
                    vals = _jq_eval(prog, doc)
                except TypeError:
                    vals = []
                vals = [x_ for x_ in vals if x_ is not None]
                if vals:
                    res = vals
                    break
            if not res:
                out.append(None)
            elif len(res) == 1:
                r = res[0]
                out.append(r if isinstance(r, str) else _json.dumps(r))
            else:
                out.append(_json.dumps(res))
        r = Series.from_pylist(s.name, out, DataType.string())
        return r.to(s.device) if s.is_gpu() else r

    return Expression(ScalarFn("jq", run, [_to_node(x)],
                               DataType.string()))
