"""String free functions beyond the str namespace (capability of
daft/functions/str_.py + daft-functions-utf8: case conversions, edit
distances, phonetics, formatting).  Host-side vectorized (these are cold
paths next to the HIP predicate/LIKE kernels in csrc/strings.hip)."""
from __future__ import annotations

import re as _re
from typing import List

from ..expressions.expressions import Expression, ScalarFn, _to_node
from ..schema import DataType
from ..series import Series


def _e(x) -> Expression:
    return x if isinstance(x, Expression) else Expression(_to_node(x))


def _host_map(name, fn, out_dt=None):
    """Build a free function applying `fn` per row on host."""
    def make(expr, *args):
        def run(s: Series, *extra) -> Series:
            vals = s.cpu().to_pylist()
            out = [None if v is None else fn(v, *extra) for v in vals]
            r = Series.from_pylist(s.name, out, out_dt or DataType.string())
            return r.to(s.device) if s.is_gpu() else r
        return Expression(ScalarFn(name, run, [_to_node(expr)],
                                   out_dt or DataType.string(),
                                   tuple(args)))
    make.__name__ = name
    return make


def _host_map2(name, fn, out_dt):
    def make(a, b, *args):
        def run(x: Series, y: Series, *extra) -> Series:
            xv = x.cpu().to_pylist()
            yv = y.cpu().to_pylist()
            n = max(len(xv), len(yv))
            out = []
            for i in range(n):
                vx = xv[i % len(xv)]
                vy = yv[i % len(yv)]
                out.append(None if vx is None or vy is None
                           else fn(vx, vy, *extra))
            r = Series.from_pylist(x.name, out, out_dt)
            return r.to(x.device) if x.is_gpu() else r
        return Expression(ScalarFn(name, run,
                                   [_to_node(a), _to_node(b)], out_dt,
                                   tuple(args)))
    make.__name__ = name
    return make


# -- case conversions -------------------------------------------------------

def _words(v: str) -> List[str]:
    parts = _re.split(r"[\s_\-]+", v)
    out: List[str] = []
    for p in parts:
        out.extend(w for w in _re.findall(
            r"[A-Z]+(?![a-z])|[A-Z][a-z]*|[a-z0-9]+", p) if w)
    return out


to_snake_case = _host_map(
    "to_snake_case", lambda v: "_".join(w.lower() for w in _words(v)))
to_upper_snake_case = _host_map(
    "to_upper_snake_case", lambda v: "_".join(w.upper() for w in _words(v)))
to_kebab_case = _host_map(
    "to_kebab_case", lambda v: "-".join(w.lower() for w in _words(v)))
to_upper_kebab_case = _host_map(
    "to_upper_kebab_case", lambda v: "-".join(w.upper() for w in _words(v)))
to_camel_case = _host_map(
    "to_camel_case",
    lambda v: "".join(w.capitalize() if i else w.lower()
                      for i, w in enumerate(_words(v))))
to_upper_camel_case = _host_map(
    "to_upper_camel_case",
    lambda v: "".join(w.capitalize() for w in _words(v)))
to_title_case = _host_map(
    "to_title_case", lambda v: " ".join(w.capitalize() for w in _words(v)))


def normalize(expr, *, remove_punct: bool = False, lowercase: bool = True,
              nfd_unicode: bool = True, white_space: bool = True):
    """Text normalization (ref: daft-functions-utf8 normalize)."""
    import unicodedata

    def one(v):
        if nfd_unicode:
            v = unicodedata.normalize("NFD", v)
            v = "".join(c for c in v if not unicodedata.combining(c))
        if lowercase:
            v = v.lower()
        if remove_punct:
            v = _re.sub(r"[^\w\s]", "", v)
        if white_space:
            v = " ".join(v.split())
        return v
    return _host_map("normalize", one)(expr)


# -- formatting -------------------------------------------------------------

def concat_ws(sep: str, *exprs):
    """Concatenate with a separator, skipping nulls (SQL CONCAT_WS)."""
    def run(*series) -> Series:
        cols = [s.cpu().to_pylist() for s in series]
        n = max(len(c) for c in cols)
        out = []
        for i in range(n):
            parts = [str(c[i % len(c)]) for c in cols
                     if c[i % len(c)] is not None]
            out.append(sep.join(parts))
        r = Series.from_pylist(series[0].name, out, DataType.string())
        return r.to(series[0].device) if series[0].is_gpu() else r
    return Expression(ScalarFn("concat_ws", run,
                               [_to_node(e) for e in exprs],
                               DataType.string()))


def format(fmt: str, *exprs):
    """Python-style {} formatting over columns (ref:
    daft/functions/str_.py format)."""
    def run(*series) -> Series:
        cols = [s.cpu().to_pylist() for s in series]
        n = max(len(c) for c in cols) if cols else 0
        out = []
        for i in range(n):
            vals = [c[i % len(c)] for c in cols]
            out.append(None if any(v is None for v in vals)
                       else fmt.format(*vals))
        r = Series.from_pylist(series[0].name if series else "fmt", out,
                               DataType.string())
        return r.to(series[0].device) if series and series[0].is_gpu() \
            else r
    return Expression(ScalarFn("format", run,
                               [_to_node(e) for e in exprs],
                               DataType.string()))


ascii_func = _host_map("ascii", lambda v: ord(v[0]) if v else 0,
                       DataType.int32())
chr_func = _host_map("chr", lambda v: chr(int(v)), DataType.string())


def space(expr):
    return _host_map("space", lambda v: " " * int(v))(expr)


def translate(expr, src: str, dst: str):
    return _host_map("translate",
                     lambda v, s, d: v.translate(str.maketrans(s, d)))(
        expr, src, dst)


def replace(expr, search: str, replacement: str):
    return _host_map("replace",
                     lambda v, a, b: v.replace(a, b))(expr, search,
                                                      replacement)


def split_part(expr, delim: str, n: int):
    def one(v, d, k):
        parts = v.split(d)
        idx = k - 1 if k > 0 else len(parts) + k
        return parts[idx] if 0 <= idx < len(parts) else ""
    return _host_map("split_part", one)(expr, delim, n)


def substring_index(expr, delim: str, n: int):
    def one(v, d, k):
        parts = v.split(d)
        if k > 0:
            return d.join(parts[:k])
        return d.join(parts[k:])
    return _host_map("substring_index", one)(expr, delim, n)


def count_matches(expr, patterns, whole_words: bool = False,
                  case_sensitive: bool = True):
    if isinstance(patterns, str):
        patterns = [patterns]
    flags = 0 if case_sensitive else _re.IGNORECASE
    pats = [(_re.compile(r"\b" + _re.escape(p) + r"\b", flags)
             if whole_words else _re.compile(_re.escape(p), flags))
            for p in patterns]

    def one(v):
        return sum(len(p.findall(v)) for p in pats)
    return _host_map("count_matches", one, DataType.int64())(expr)


# -- regex ------------------------------------------------------------------

def regexp(expr, pattern: str):
    return _e(expr).str.match(pattern)


def regexp_count(expr, pattern: str):
    p = _re.compile(pattern)
    return _host_map("regexp_count", lambda v: len(p.findall(v)),
                     DataType.int64())(expr)


def regexp_extract(expr, pattern: str, group: int = 0):
    p = _re.compile(pattern)

    def one(v):
        m = p.search(v)
        return m.group(group) if m else None
    return _host_map("regexp_extract", one)(expr)


def regexp_extract_all(expr, pattern: str, group: int = 0):
    p = _re.compile(pattern)

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else
               [(m.group(group)) for m in p.finditer(v)] for v in vals]
        r = Series.from_pylist(s.name, out,
                               DataType.list(DataType.string()))
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("regexp_extract_all", run,
                               [_to_node(expr)],
                               DataType.list(DataType.string())))


def regexp_replace(expr, pattern: str, replacement: str):
    p = _re.compile(pattern)
    return _host_map("regexp_replace",
                     lambda v: p.sub(replacement, v))(expr)


def regexp_split(expr, pattern: str):
    p = _re.compile(pattern)

    def run(s: Series) -> Series:
        vals = s.cpu().to_pylist()
        out = [None if v is None else p.split(v) for v in vals]
        r = Series.from_pylist(s.name, out,
                               DataType.list(DataType.string()))
        return r.to(s.device) if s.is_gpu() else r
    return Expression(ScalarFn("regexp_split", run, [_to_node(expr)],
                               DataType.list(DataType.string())))


# -- distances / phonetics --------------------------------------------------

def _lev(a: str, b: str) -> int:
    if len(a) < len(b):
        a, b = b, a
    prev = list(range(len(b) + 1))
    for i, ca in enumerate(a, 1):
        cur = [i]
        for j, cb in enumerate(b, 1):
            cur.append(min(prev[j] + 1, cur[j - 1] + 1,
                           prev[j - 1] + (ca != cb)))
        prev = cur
    return prev[-1]


def _lev_series(x: Series, y: Series) -> Series:
    """GPU path: thread-per-pair HIP DP kernel over ASCII rows; rows the
    kernel flags (-1: non-ASCII or longer than 512 bytes) recompute on
    the host character-level oracle."""
    import torch as _t
    from ..kernels import load_native
    nat = load_native()
    gpu_ok = (x.is_gpu() and y.is_gpu() and nat is not None and
              len(x) == len(y) and not x.is_dict() and not y.is_dict()
              and x.offsets is not None and y.offsets is not None)
    if gpu_ok:
        MAXL, CHUNK = 512, 1 << 20
        outs = []
        n = len(x)
        for lo in range(0, n, CHUNK):
            xs = x.slice(lo, lo + CHUNK)
            ys = y.slice(lo, lo + CHUNK)
            outs.append(nat.levenshtein(xs.offsets, xs.data, ys.offsets,
                                        ys.data, MAXL))
        d = _t.cat(outs).to(_t.int64)
        bad = (d < 0).nonzero().reshape(-1)
        validity = None
        if x.validity is not None or y.validity is not None:
            xv = x.validity if x.validity is not None else                 _t.ones(n, dtype=_t.bool, device=x.device)
            yv = y.validity if y.validity is not None else                 _t.ones(n, dtype=_t.bool, device=y.device)
            validity = xv & yv
        res = Series(x.name, DataType.int64(), data=d, validity=validity)
        if int(bad.numel()):
            xs = x.take(bad).cpu().to_pylist()
            ys = y.take(bad).cpu().to_pylist()
            fix = _t.tensor([0 if (a is None or b is None)
                             else _lev(a, b) for a, b in zip(xs, ys)],
                            dtype=_t.int64, device=x.device)
            d = d.clone()
            d[bad] = fix
            res = Series(x.name, DataType.int64(), data=d,
                         validity=validity)
        return res
    xv = x.cpu().to_pylist()
    yv = y.cpu().to_pylist()
    n = max(len(xv), len(yv))
    out = [None if xv[i % len(xv)] is None or yv[i % len(yv)] is None
           else _lev(xv[i % len(xv)], yv[i % len(yv)]) for i in range(n)]
    r = Series.from_pylist(x.name, out, DataType.int64())
    return r.to(x.device) if x.is_gpu() else r


def levenshtein_distance(a, b):
    return Expression(ScalarFn("levenshtein_distance", _lev_series,
                               [_to_node(a), _to_node(b)],
                               DataType.int64()))


def _dlev(a: str, b: str) -> int:
    d = {}
    la, lb = len(a), len(b)
    for i in range(la + 1):
        d[i, 0] = i
    for j in range(lb + 1):
        d[0, j] = j
    for i in range(1, la + 1):
        for j in range(1, lb + 1):
            cost = a[i - 1] != b[j - 1]
            d[i, j] = min(d[i - 1, j] + 1, d[i, j - 1] + 1,
                          d[i - 1, j - 1] + cost)
            if i > 1 and j > 1 and a[i - 1] == b[j - 2] and \
                    a[i - 2] == b[j - 1]:
                d[i, j] = min(d[i, j], d[i - 2, j - 2] + 1)
    return d[la, lb]


damerau_levenshtein_distance = _host_map2("damerau_levenshtein_distance",
                                          _dlev, DataType.int64())


def _ham_str(a: str, b: str) -> int:
    return sum(x != y for x, y in zip(a, b)) + abs(len(a) - len(b))


hamming_distance_str = _host_map2("hamming_distance_str", _ham_str,
                                  DataType.int64())


def _jaro(a: str, b: str) -> float:
    if a == b:
        return 1.0
    la, lb = len(a), len(b)
    if not la or not lb:
        return 0.0
    win = max(la, lb) // 2 - 1
    ma = [False] * la
    mb = [False] * lb
    matches = 0
    for i, ca in enumerate(a):
        lo = max(0, i - win)
        hi = min(lb, i + win + 1)
        for j in range(lo, hi):
            if not mb[j] and b[j] == ca:
                ma[i] = mb[j] = True
                matches += 1
                break
    if not matches:
        return 0.0
    t = 0
    k = 0
    for i in range(la):
        if ma[i]:
            while not mb[k]:
                k += 1
            if a[i] != b[k]:
                t += 1
            k += 1
    t //= 2
    return (matches / la + matches / lb + (matches - t) / matches) / 3


jaro_similarity = _host_map2("jaro_similarity", _jaro, DataType.float64())


def _jaro_winkler(a: str, b: str) -> float:
    j = _jaro(a, b)
    pre = 0
    for x, y in zip(a[:4], b[:4]):
        if x == y:
            pre += 1
        else:
            break
    return j + pre * 0.1 * (1 - j)


jaro_winkler_similarity = _host_map2("jaro_winkler_similarity",
                                     _jaro_winkler, DataType.float64())


def _jaccard(a: str, b: str) -> float:
    sa, sb = set(a), set(b)
    if not sa and not sb:
        return 1.0
    return len(sa & sb) / len(sa | sb)


jaccard_similarity = _host_map2("jaccard_similarity", _jaccard,
                                DataType.float64())

_SOUNDEX = {**{c: "1" for c in "bfpv"}, **{c: "2" for c in "cgjkqsxz"},
            **{c: "3" for c in "dt"}, "l": "4",
            **{c: "5" for c in "mn"}, "r": "6"}


def _soundex(v: str) -> str:
    v = "".join(c for c in v.lower() if c.isalpha())
    if not v:
        return ""
    out = v[0].upper()
    prev = _SOUNDEX.get(v[0], "")
    for c in v[1:]:
        code = _SOUNDEX.get(c, "")
        if code and code != prev:
            out += code
        if c not in "hw":
            prev = code
        if len(out) == 4:
            break
    return (out + "000")[:4]


soundex = _host_map("soundex", _soundex)
