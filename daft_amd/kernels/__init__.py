"""Kernel dispatch: GPU tensors -> hand-written HIP/CDNA4 extension
(`daft_amd._native`, built from csrc/ for gfx950); CPU tensors -> torch/numpy
fallbacks (test path only — there is no GPU in CI).

This is the seam the reference implements as daft-core's Rust kernels
(/root/reference/src/daft-core/src/kernels/, src/array/ops/).  On a GPU box
the native extension is REQUIRED: any GPU-path call raises if the extension
failed to load, so a silent eager fallback cannot masquerade as the HIP path.
"""
from __future__ import annotations

from typing import List, Optional, Sequence

import numpy as np
import torch

from ..schema import DataType, TypeKind, supertype
from ..series import Series, full_null

_native = None
_native_err: Optional[str] = None
_SIGNFLIP64 = -0x8000000000000000  # order-preserving u64 -> i64 bit flip


def load_native():
    """Load the in-tree HIP extension.  Returns the module or None (CPU-only)."""
    global _native, _native_err
    if _native is not None or _native_err is not None:
        return _native
    try:
        from daft_amd import _native as mod  # built by setup.py build_ext --inplace
        _native = mod
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _native_err = str(e)
        _native = None
    return _native


def native_required():
    """GPU path: return the extension, raising loudly if it is missing."""
    mod = load_native()
    if mod is None:
        raise RuntimeError(
            "daft_amd HIP extension (_native) is not built but a GPU code "
            f"path was hit. Build with `python setup.py build_ext --inplace` "
            f"(PYTORCH_ROCM_ARCH=gfx950). Import error: {_native_err}")
    return mod


# ---------------------------------------------------------------------------
# column descriptors for generic row ops (hash / compare) on the GPU
# ---------------------------------------------------------------------------
# tags shared with csrc/row_ops.h
_TAG_W1, _TAG_W2, _TAG_W4, _TAG_W8, _TAG_STR = 0, 1, 2, 3, 4
_TAG_F32, _TAG_F64 = 5, 6


def _col_desc(s: Series):
    """(tag, data_tensor, offsets_or_none, validity_or_none) for row ops."""
    k = s.dtype.kind
    if s.is_dict():
        # codes are equality-preserving within one column's vocab
        return (_TAG_W4, s.data, None, s.validity)
    if k in (TypeKind.STRING, TypeKind.BINARY):
        return (_TAG_STR, s.data, s.offsets, s.validity)
    data = s.data
    if data is None:
        raise TypeError(f"column {s.name}: {s.dtype!r} not usable as row key")
    if data.dtype == torch.float32:
        return (_TAG_F32, data, None, s.validity)
    if data.dtype == torch.float64:
        return (_TAG_F64, data, None, s.validity)
    width = data.element_size()
    tag = {1: _TAG_W1, 2: _TAG_W2, 4: _TAG_W4, 8: _TAG_W8}[width]
    return (tag, data, None, s.validity)


def _descs(series: Sequence[Series]):
    tags, datas, offs, vals = [], [], [], []
    for s in series:
        if s.dtype.is_decimal() and s.children:
            raise NotImplementedError(
                "GPU group/join/partition keys of wide decimals (p>18) "
                "are not supported — cast to a p<=18 decimal, integer or "
                "string key first (CPU grouping of wide keys works via "
                "the exact host path)")
        t, d, o, v = _col_desc(s)
        tags.append(t)
        datas.append(d)
        offs.append(o)
        vals.append(v)
    return tags, datas, offs, vals


def _is_gpu(series_or_tensor) -> bool:
    if isinstance(series_or_tensor, Series):
        return series_or_tensor.is_gpu()
    return series_or_tensor.is_cuda


# ---------------------------------------------------------------------------
# take / filter / concat
# ---------------------------------------------------------------------------

def compact_indices(mask: Series) -> torch.Tensor:
    """Boolean mask (+validity: null = drop) -> selected row indices."""
    m = mask.data
    if mask.validity is not None:
        m = m & mask.validity
    if _is_gpu(m):
        return native_required().compact_indices(m)
    return torch.nonzero(m, as_tuple=False).reshape(-1).to(torch.int64)


def take(s: Series, indices: torch.Tensor,
         has_neg: Optional[bool] = None) -> Series:
    """Gather rows; index -1 yields null.  `has_neg` lets multi-column
    callers hoist the device sync for the negative-index check."""
    k = s.dtype.kind
    n_out = int(indices.shape[0])
    dev = s.device
    indices = indices.to(dev)
    if has_neg is None:
        has_neg = bool((indices < 0).any().item()) if n_out else False
    safe_idx = indices.clamp(min=0) if has_neg else indices

    validity = None
    if s.validity is not None:
        validity = s.validity[safe_idx]
    if has_neg:
        pos = indices >= 0
        validity = pos if validity is None else (validity & pos)

    if k == TypeKind.PYTHON:
        idx_cpu = indices.cpu().numpy()
        objs = [None if i < 0 else s.pyobjs[i] for i in idx_cpu]
        return Series(s.name, s.dtype, pyobjs=objs, validity=validity,
                      length=n_out)
    if s.is_dict():
        codes = s.data[safe_idx] if n_out else s.data[:0]
        return Series.make_dict(s.name, s.children[0], codes, validity)
    if k in (TypeKind.STRING, TypeKind.BINARY):
        if _is_gpu(s):
            new_off, new_bytes = native_required().take_string(
                s.offsets, s.data, safe_idx)
        else:
            new_off, new_bytes = _cpu_take_string(s.offsets, s.data, safe_idx)
        return Series(s.name, s.dtype, data=new_bytes, offsets=new_off,
                      validity=validity)
    if k in (TypeKind.LIST, TypeKind.MAP):
        # gather child ranges
        lens = s.offsets[1:] - s.offsets[:-1]
        sel_lens = lens[safe_idx]
        new_off = torch.zeros(n_out + 1, dtype=torch.int64, device=dev)
        torch.cumsum(sel_lens, 0, out=new_off[1:])
        starts = s.offsets[:-1][safe_idx]
        child_idx = _expand_ranges(starts, sel_lens, new_off)
        child = s.children[0].take(child_idx)
        return Series(s.name, s.dtype, offsets=new_off, children=[child],
                      validity=validity)
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING,
             TypeKind.FIXED_SHAPE_TENSOR):
        if k == TypeKind.FIXED_SHAPE_TENSOR:
            sz = 1
            for d in s.dtype.shape:
                sz *= d
        else:
            sz = s.dtype.size
        child_idx = (safe_idx.unsqueeze(1) * sz +
                     torch.arange(sz, device=dev)).reshape(-1)
        child = s.children[0].take(child_idx)
        return Series(s.name, s.dtype, children=[child], validity=validity,
                      length=n_out)
    if k == TypeKind.STRUCT or (k == TypeKind.DECIMAL128 and s.children):
        children = [c.take(safe_idx, has_neg=False) for c in s.children]
        return Series(s.name, s.dtype, children=children, validity=validity,
                      length=n_out)
    # fixed width: torch gather (hipified index_select is memory-bound optimal
    # for contiguous gathers; a fused HIP gather handles multi-column takes
    # at the recordbatch layer).  Unsigned ints gather through a signed view
    # (no unsigned index_cuda kernels in torch-rocm).
    d = s.data
    _signed = {torch.uint16: torch.int16, torch.uint32: torch.int32,
               torch.uint64: torch.int64}
    uview = _signed.get(d.dtype)
    if uview is not None:
        d = d.view(uview)
    data = d[safe_idx] if n_out else d[:0]
    if uview is not None:
        data = data.view(s.data.dtype)
    return Series(s.name, s.dtype, data=data, validity=validity)


def _expand_ranges(starts: torch.Tensor, lens: torch.Tensor,
                   out_offsets: torch.Tensor) -> torch.Tensor:
    """child indices for gathered list ranges: concat(arange(start_i, start_i+len_i))."""
    total = int(out_offsets[-1].item()) if out_offsets.numel() > 1 else 0
    dev = starts.device
    if total == 0:
        return torch.zeros(0, dtype=torch.int64, device=dev)
    # position j in output belongs to row i where out_offsets[i] <= j < out_offsets[i+1]
    j = torch.arange(total, dtype=torch.int64, device=dev)
    row = torch.searchsorted(out_offsets[1:], j, right=True)
    return starts[row] + (j - out_offsets[row])


def _cpu_take_string(offsets: torch.Tensor, data: torch.Tensor,
                     idx: torch.Tensor):
    off = offsets.numpy()
    idx_np = idx.numpy()
    lens = (off[1:] - off[:-1])[idx_np] if len(idx_np) else np.zeros(0, np.int64)
    new_off = np.zeros(len(idx_np) + 1, dtype=np.int64)
    np.cumsum(lens, out=new_off[1:])
    total = int(new_off[-1])
    out = np.zeros(total, dtype=np.uint8)
    src = data.numpy()
    for i, r in enumerate(idx_np):
        a, b = new_off[i], new_off[i + 1]
        out[a:b] = src[off[r]:off[r] + (b - a)]
    return torch.from_numpy(new_off), torch.from_numpy(out)


def concat(series: List[Series]) -> Series:
    assert series, "concat of zero series"
    if len(series) == 1:
        return series[0]
    s0 = series[0]
    dtype = s0.dtype
    for s in series[1:]:
        if s.dtype != dtype:
            dtype = supertype(dtype, s.dtype)
    series = [s.cast(dtype) if s.dtype != dtype else s for s in series]
    dev = s0.device
    total = sum(len(s) for s in series)
    k = dtype.kind

    validity = None
    if any(s.validity is not None for s in series):
        parts = [s.validity if s.validity is not None else
                 torch.ones(len(s), dtype=torch.bool, device=dev)
                 for s in series]
        validity = torch.cat(parts)

    if k == TypeKind.PYTHON:
        objs = []
        for s in series:
            objs.extend(s.pyobjs)
        return Series(s0.name, dtype, pyobjs=objs, validity=validity,
                      length=total)
    if any(s.is_dict() for s in series):
        if all(s.is_dict() for s in series) and \
                all(_same_vocab(s.children[0], series[0].children[0])
                    for s in series[1:]):
            codes = torch.cat([s.data for s in series])
            return Series.make_dict(s0.name, series[0].children[0], codes,
                                    validity)
        series = [s.dict_decode() for s in series]
        data = torch.cat([s.data for s in series])
        new_off = torch.zeros(total + 1, dtype=torch.int64, device=dev)
        pos, base = 0, 0
        for s in series:
            n = len(s)
            new_off[pos + 1: pos + n + 1] = s.offsets[1:] + base
            base += int(s.offsets[-1].item())
            pos += n
        return Series(s0.name, dtype, data=data, offsets=new_off,
                      validity=validity)
    if k in (TypeKind.STRING, TypeKind.BINARY, TypeKind.LIST,
             TypeKind.MAP):
        new_bytes_parts = []
        new_off = torch.zeros(total + 1, dtype=torch.int64, device=dev)
        pos = 0
        base = 0
        for s in series:
            n = len(s)
            new_off[pos + 1: pos + n + 1] = s.offsets[1:] + base
            base += int(s.offsets[-1].item())
            pos += n
        if k in (TypeKind.LIST, TypeKind.MAP):
            child = concat([s.children[0] for s in series])
            return Series(s0.name, dtype, offsets=new_off, children=[child],
                          validity=validity)
        data = torch.cat([s.data for s in series])
        return Series(s0.name, dtype, data=data, offsets=new_off,
                      validity=validity)
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING,
             TypeKind.FIXED_SHAPE_TENSOR, TypeKind.STRUCT) or \
            (k == TypeKind.DECIMAL128 and s0.children):
        nchild = len(s0.children)
        children = [concat([s.children[i] for s in series])
                    for i in range(nchild)]
        return Series(s0.name, dtype, children=children, validity=validity,
                      length=total)
    data = torch.cat([s.data for s in series])
    return Series(s0.name, dtype, data=data, validity=validity)


# ---------------------------------------------------------------------------
# casts
# ---------------------------------------------------------------------------

def cast(s: Series, dtype: DataType) -> Series:
    if s.dtype == dtype:
        return s
    if s.is_dict():
        s = s.dict_decode()
        if s.dtype == dtype:
            return s
    k, nk = s.dtype.kind, dtype.kind
    if nk == TypeKind.NULL:
        return full_null(s.name, dtype, len(s), s.device)
    if k == TypeKind.NULL:
        return full_null(s.name, dtype, len(s), s.device)
    if k == TypeKind.DECIMAL128 and s.children:
        # wide (p>18) source: limb arithmetic (kernels/decimal128.py)
        from . import decimal128 as d128
        lo, hi = d128.limbs(s)
        sc = s.dtype.scale
        if nk == TypeKind.DECIMAL128:
            if dtype.scale > sc:
                lo, hi = d128.mul128_pow10(lo, hi, dtype.scale - sc)
            elif dtype.scale < sc:
                lo, hi = d128.divround128_pow10(lo, hi, sc - dtype.scale)
            if dtype.precision > 18:
                return d128.make(s.name, dtype, lo, hi, s.validity)
            # narrowing: must fit a scaled int64
            if bool((hi != (lo >> 63)).any()):
                raise ValueError(
                    f"decimal cast overflow: {s.dtype!r} -> {dtype!r}")
            return Series(s.name, dtype, data=lo, validity=s.validity)
        if dtype.is_floating():
            out = (d128.to_float64(lo, hi) / (10.0 ** sc)) \
                .to(dtype.to_torch())
            return Series(s.name, dtype, data=out, validity=s.validity)
        if dtype.is_integer():
            ilo, _ihi = d128.divround128_pow10(lo, hi, sc,
                                               round_half=False) \
                if sc else (lo, hi)
            tdt = dtype.to_torch()
            out = ilo if tdt == torch.int64 else (
                ilo.view(torch.uint64) if tdt == torch.uint64
                else ilo.to(tdt))
            return Series(s.name, dtype, data=out, validity=s.validity)
        # strings / everything else: exact host path below (to_pylist)
    if nk == TypeKind.DECIMAL128 and dtype.precision > 18 and \
            s.dtype.is_numeric() and (s.data is not None or s.children):
        from . import decimal128 as d128
        if k == TypeKind.DECIMAL128:      # narrow -> wide: sign-extend
            lo, hi = d128.from_int64(s.data)
            dsc = dtype.scale - s.dtype.scale
        elif s.dtype.is_floating():
            v = s.data.to(torch.float64) * (10.0 ** dtype.scale)
            fhi = torch.floor(v / float(1 << 64))
            flo = v - fhi * float(1 << 64)       # in [0, 2^64)
            half = float(1 << 63)
            lo = torch.where(flo >= half, (flo - half).to(torch.int64)
                             + (-(1 << 63)), flo.to(torch.int64))
            hi = fhi.to(torch.int64)
            dsc = 0
        else:
            lo, hi = d128.from_int64(s.data.to(torch.int64))
            dsc = dtype.scale
        if dsc > 0:
            lo, hi = d128.mul128_pow10(lo, hi, dsc)
        elif dsc < 0:
            lo, hi = d128.divround128_pow10(lo, hi, -dsc)
        return d128.make(s.name, dtype, lo, hi, s.validity)
    if k == TypeKind.DECIMAL128 and s.data is not None and \
            s.data.dtype == torch.int64:
        sc = s.dtype.scale
        if nk == TypeKind.DECIMAL128:
            if dtype.to_physical().kind == TypeKind.INT64:
                d = s.data
                if dtype.scale > sc:
                    d = d * (10 ** (dtype.scale - sc))
                elif dtype.scale < sc:
                    div = 10 ** (sc - dtype.scale)
                    d = torch.div(d + torch.sign(d) * (div // 2), div,
                                  rounding_mode="trunc")
                return Series(s.name, dtype, data=d, validity=s.validity)
            return Series(s.name, dtype,
                          data=s.data.to(torch.float64) / (10 ** sc),
                          validity=s.validity)
        if dtype.is_floating():
            out = (s.data.to(torch.float64) / (10 ** sc)) \
                .to(dtype.to_torch())
            return Series(s.name, dtype, data=out, validity=s.validity)
        if dtype.is_integer():
            out = torch.div(s.data, 10 ** sc, rounding_mode="trunc") \
                .to(dtype.to_torch())
            return Series(s.name, dtype, data=out, validity=s.validity)
    if nk == TypeKind.DECIMAL128 and \
            dtype.to_physical().kind == TypeKind.INT64 and \
            s.data is not None and s.dtype.is_numeric():
        mul = 10 ** dtype.scale
        if s.dtype.is_floating() or (s.dtype.is_decimal() and
                                     s.data.dtype == torch.float64):
            out = torch.round(s.data.to(torch.float64) * mul) \
                .to(torch.int64)
        else:
            out = s.data.to(torch.int64) * mul
        return Series(s.name, dtype, data=out, validity=s.validity)
    if dtype.is_fixed_width() and s.dtype.is_fixed_width():
        tdt = dtype.to_torch()
        data = s.data
        if k == TypeKind.BOOL and nk != TypeKind.BOOL:
            data = data.to(torch.int8)
        if nk == TypeKind.BOOL:
            out = data != 0
        else:
            out = data.to(tdt)
        return Series(s.name, dtype, data=out, validity=s.validity)
    if nk == TypeKind.STRING:
        # host-side stringification (display path, not hot)
        vals = s.to_pylist()
        out = [None if v is None else _to_str(v) for v in vals]
        return Series.from_pylist(s.name, out, DataType.string(),
                                  device=s.device)
    if k == TypeKind.STRING and dtype.is_numeric():
        vals = s.to_pylist()
        if dtype.is_decimal():
            conv = lambda x: x  # Decimal(str) parse in from_pylist: exact
        elif dtype.is_floating():
            conv = float
        else:
            conv = int
        out = [None if v is None or v == "" else conv(v) for v in vals]
        return Series.from_pylist(s.name, out, dtype, device=s.device)
    if k == TypeKind.STRING and nk == TypeKind.DATE:
        vals = s.to_pylist()
        out = [None if v is None else _dt_parse_date(v) for v in vals]
        return Series.from_pylist(s.name, out, dtype, device=s.device)
    if k == TypeKind.LIST and nk == TypeKind.LIST:
        child = s.children[0].cast(dtype.inner)
        return Series(s.name, dtype, offsets=s.offsets, children=[child],
                      validity=s.validity)
    if (k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING) and
            nk in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING)):
        child = s.children[0].cast(dtype.inner)
        return Series(s.name, dtype, children=[child], validity=s.validity,
                      length=len(s))
    if k == TypeKind.LIST and nk in (TypeKind.FIXED_SIZE_LIST,
                                     TypeKind.EMBEDDING,
                                     TypeKind.FIXED_SHAPE_TENSOR):
        if nk == TypeKind.FIXED_SHAPE_TENSOR:
            size = 1
            for d in dtype.shape:
                size *= d
            inner = dtype.inner
        else:
            size, inner = dtype.size, dtype.inner
        lens = s.offsets[1:] - s.offsets[:-1]
        ok = lens == size
        if s.validity is not None:
            ok = ok | ~s.validity
        if not bool(ok.all().item()):
            raise ValueError(
                f"cannot cast ragged list to {dtype!r}: lengths differ")
        # gather child rows densely (null rows fill with nulls)
        idx = (s.offsets[:-1].unsqueeze(1) +
               torch.arange(size, device=s.device)).reshape(-1)
        if s.validity is not None:
            idx = torch.where(
                s.validity.repeat_interleave(size), idx,
                torch.full_like(idx, -1))
        child = s.children[0].take(idx).cast(inner)
        return Series(s.name, dtype, children=[child], validity=s.validity,
                      length=len(s))
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING) and \
            nk == TypeKind.LIST:
        offs = torch.arange(0, (len(s) + 1) * s.dtype.size, s.dtype.size,
                            dtype=torch.int64, device=s.device)
        child = s.children[0].cast(dtype.inner)
        return Series(s.name, dtype, offsets=offs, children=[child],
                      validity=s.validity)
    raise TypeError(f"unsupported cast {s.dtype!r} -> {dtype!r}")


def _to_str(v) -> str:
    if isinstance(v, float):
        return repr(v)
    if isinstance(v, bytes):
        return v.decode("utf-8", "replace")
    return str(v)


def _dt_parse_date(v: str):
    import datetime as dtmod
    return dtmod.date.fromisoformat(v)


# ---------------------------------------------------------------------------
# elementwise binary / compare / logical
# ---------------------------------------------------------------------------

_COMPARE_TORCH = {"eq": torch.eq, "ne": torch.ne, "lt": torch.lt,
                  "le": torch.le, "gt": torch.gt, "ge": torch.ge}


def _align(l: Series, r: Series):
    """Length-align operands.  Length-1 fixed-width operands stay length-1
    (torch elementwise ops broadcast [1] against [n] — materializing the
    literal into a full column costs a pointless HBM round trip)."""
    if len(l) == len(r):
        return l, r
    if len(l) == 1:
        if l.dtype.is_fixed_width() and l.data is not None:
            return l, r
        return l.broadcast(len(r)), r
    if len(r) == 1:
        if r.dtype.is_fixed_width() and r.data is not None:
            return l, r
        return l, r.broadcast(len(l))
    raise ValueError(f"length mismatch {len(l)} vs {len(r)}")



def _expand_validity(validity, n: int):
    """Broadcast a length-1 validity to the output length (null literal)."""
    if validity is not None and validity.numel() == 1 and n != 1:
        return validity.expand(n).contiguous()
    return validity


def binary_op(l: Series, r: Series, op: str) -> Series:
    l, r = _align(l, r)
    validity = _null_validity(l, r)
    if l.dtype.is_temporal() or r.dtype.is_temporal():
        return _temporal_binary(l, r, op, validity)
    if (l.dtype.is_decimal() or r.dtype.is_decimal()):
        out = _decimal_binary(l, r, op, validity)
        if out is not None:
            return out
    out_dt = supertype(l.dtype, r.dtype)
    if op == "div" and not out_dt.is_floating():
        out_dt = DataType.float64()
    tdt = out_dt.to_torch()
    a = l.data.to(tdt)
    b = r.data.to(tdt)
    if op == "add":
        out = a + b
    elif op == "sub":
        out = a - b
    elif op == "mul":
        out = a * b
    elif op == "div":
        out = a / b
    elif op == "floordiv":
        out = torch.div(a, b, rounding_mode="floor")
    elif op == "mod":
        out = torch.remainder(a, b)
    elif op == "pow":
        out = torch.pow(a, b)
    else:
        raise ValueError(f"unknown binary op {op}")
    return Series(l.name, out_dt, data=out,
                  validity=_expand_validity(validity, int(out.shape[0])))


def _dec_phys_int(dt: DataType) -> bool:
    return dt.is_decimal() and dt.to_physical().kind == TypeKind.INT64


def _decimal_binary(l: Series, r: Series, op: str, validity):
    """Exact decimal arithmetic on scaled int64 (ref semantics:
    daft-core decimal ops).  add/sub align scales; mul adds scales;
    div (and anything overflowing p=18) promotes to float64.  Returns
    None to fall through to the float path."""
    def to_float(s: Series) -> Series:
        return cast(s, DataType.float64()) if s.dtype.is_decimal() else s

    from . import decimal128 as d128
    any_wide = (l.dtype.is_decimal() and d128.is_wide(l.dtype)) or \
        (r.dtype.is_decimal() and d128.is_wide(r.dtype))
    if any_wide:
        if op in ("add", "sub", "mul") and not l.dtype.is_floating() \
                and not r.dtype.is_floating():
            return _decimal_wide_binary(l, r, op, validity)
        return binary_op(to_float(l), to_float(r), op)   # div etc: f64

    if not (_dec_phys_int(l.dtype) or _dec_phys_int(r.dtype)) or \
            l.dtype.is_floating() or r.dtype.is_floating() or \
            op in ("div", "pow"):
        ll, rr = to_float(l), to_float(r)
        if ll is l and rr is r:
            return None
        return binary_op(ll, rr, op)

    def scaled(s: Series):
        if s.dtype.is_decimal():
            return s.data.to(torch.int64), s.dtype.scale, s.dtype.precision
        if len(s) == 1:  # literal: use its true digit count
            v = abs(int(s.data.item()))
            return s.data.to(torch.int64), 0, max(1, len(str(v)))
        return s.data.to(torch.int64), 0, 19

    av, asc, ap = scaled(l)
    bv, bsc, bp = scaled(r)
    # result dtype via the SAME rule static typing uses
    # (schema.decimal_binary_result) — schema and runtime never diverge
    from ..schema import decimal_binary_result

    def int_digits(s, p):
        return p if not s.dtype.is_decimal() else None
    res_dt = decimal_binary_result(l.dtype, r.dtype, op,
                                   int_digits(l, ap), int_digits(r, bp))
    if res_dt is None or res_dt.is_floating():
        return binary_op(to_float(l), to_float(r), op)
    if op in ("add", "sub"):
        if res_dt.precision > 18:
            # overflows scaled int64: exact two-limb wide path
            return _decimal_wide_binary(l, r, op, validity, res_dt)
        sc = res_dt.scale
        if asc < sc:
            av = av * (10 ** (sc - asc))
        if bsc < sc:
            bv = bv * (10 ** (sc - bsc))
        out = av + bv if op == "add" else av - bv
        return Series(l.name, res_dt, data=out,
                      validity=_expand_validity(validity,
                                                int(out.shape[0])))
    if op == "mul":
        if res_dt.precision > 18:
            return _decimal_wide_binary(l, r, op, validity, res_dt)
        out = av * bv
        return Series(l.name, res_dt, data=out,
                      validity=_expand_validity(validity,
                                                int(out.shape[0])))
    return None


def _decimal_wide_binary(l: Series, r: Series, op: str, validity,
                         res_dt=None):
    """Exact add/sub/mul when either operand (or the result) is a wide
    decimal: two-limb carry arithmetic (kernels/decimal128.py), device-
    portable torch ops."""
    from . import decimal128 as d128

    def parts(s: Series):
        if s.dtype.is_decimal():
            if d128.is_wide(s.dtype):
                lo, hi = d128.limbs(s)
            else:
                lo, hi = d128.from_int64(s.data.to(torch.int64))
            return lo, hi, s.dtype.scale, s.dtype.precision
        v = s.data.to(torch.int64)
        lo, hi = d128.from_int64(v)
        if len(s) == 1:
            p = max(1, len(str(abs(int(v.item())))))
        else:
            p = 19
        return lo, hi, 0, p

    alo, ahi, asc, ap = parts(l)
    blo, bhi, bsc, bp = parts(r)
    if op in ("add", "sub"):
        sc = max(asc, bsc)
        if asc < sc:
            alo, ahi = d128.mul128_pow10(alo, ahi, sc - asc)
        if bsc < sc:
            blo, bhi = d128.mul128_pow10(blo, bhi, sc - bsc)
        olo, ohi = (d128.add128 if op == "add" else d128.sub128)(
            alo, ahi, blo, bhi)
        p = min(38, max(ap - asc, bp - bsc) + sc + 1)
    else:                                   # mul
        sc = asc + bsc
        olo, ohi = d128.mul128(alo, ahi, blo, bhi)
        p = min(38, ap + bp + 1)
    dt = res_dt if res_dt is not None else \
        DataType.decimal128(max(p, 19), sc)
    return d128.make(l.name, dt, olo, ohi,
                     _expand_validity(validity, int(olo.numel())))


def _temporal_binary(l: Series, r: Series, op: str, validity):
    lk, rk = l.dtype.kind, r.dtype.kind
    if op == "sub" and lk == rk == TypeKind.DATE:
        out = (l.data.to(torch.int64) - r.data.to(torch.int64))
        return Series(l.name, DataType.duration("us"),
                      data=out * 86_400_000_000,
                      validity=_expand_validity(validity, int(out.shape[0])))
    if lk == TypeKind.DATE and r.dtype.is_integer():
        out = l.data + r.data.to(torch.int32) * (1 if op == "add" else -1)
        return Series(l.name, l.dtype, data=out,
                      validity=_expand_validity(validity, int(out.shape[0])))
    if lk == TypeKind.TIMESTAMP and rk == TypeKind.DURATION:
        out = l.data + r.data * (1 if op == "add" else -1)
        return Series(l.name, l.dtype, data=out,
                      validity=_expand_validity(validity, int(out.shape[0])))
    if op == "sub" and lk == rk == TypeKind.TIMESTAMP:
        out = l.data - r.data
        return Series(l.name, DataType.duration(l.dtype.timeunit),
                      data=out,
                      validity=_expand_validity(validity, int(out.shape[0])))
    raise TypeError(f"temporal op {op} on {l.dtype!r}, {r.dtype!r}")


def _null_validity(l: Series, r: Series):
    lv, rv = l.validity, r.validity
    if lv is None:
        return None if rv is None else rv.clone()
    if rv is None:
        return lv.clone()
    return lv & rv


def compare_op(l: Series, r: Series, op: str) -> Series:
    # dict vs scalar-literal fast path BEFORE broadcasting (avoids
    # materializing a full-length string column for the literal)
    if op in ("eq", "ne"):
        if l.is_dict() and len(r) == 1 and not r.is_dict():
            return _dict_eq_literal(l, r, op)
        if r.is_dict() and len(l) == 1 and not l.is_dict():
            return _dict_eq_literal(r, l, op)
    l, r = _align(l, r)
    validity = _null_validity(l, r)
    lk = l.dtype.kind
    if lk in (TypeKind.STRING, TypeKind.BINARY) or \
            r.dtype.kind in (TypeKind.STRING, TypeKind.BINARY):
        return _string_compare(l, r, op, validity)
    if l.dtype.is_decimal() or r.dtype.is_decimal():
        # align scales exactly (scaled-int compare), or promote to f64
        st = supertype(l.dtype, r.dtype)
        if l.dtype != st:
            l = cast(l, st)
        if r.dtype != st:
            r = cast(r, st)
        from . import decimal128 as d128
        if st.is_decimal() and d128.is_wide(st):
            out = d128.cmp128(*d128.limbs(l), *d128.limbs(r), op)
            return Series(l.name, DataType.bool(), data=out,
                          validity=validity)
    st = supertype(l.dtype, r.dtype)
    tdt = st.to_torch()

    def conv(t: torch.Tensor) -> torch.Tensor:
        # unsigned wide ints lack cuda/cpu comparison kernels in torch:
        # compare through an order-preserving signed view
        if tdt == torch.uint64:
            v = t.view(torch.int64) if t.dtype == torch.uint64 \
                else t.to(torch.int64)
            return v ^ _SIGNFLIP64
        if tdt in (torch.uint16, torch.uint32):
            src = {torch.uint16: torch.int16,
                   torch.uint32: torch.int32}.get(t.dtype)
            return (t.view(src) if src is not None else t).to(torch.int64) & \
                ((1 << (16 if tdt == torch.uint16 else 32)) - 1)
        return t.to(tdt)
    a, b = conv(l.data), conv(r.data)
    out = _COMPARE_TORCH[op](a, b)
    return Series(l.name, DataType.bool(), data=out,
                  validity=_expand_validity(validity, int(out.shape[0])))


def _same_vocab(a: Series, b: Series) -> bool:
    if a is b:
        return True
    if len(a) != len(b):
        return False
    return bool(torch.equal(a.offsets, b.offsets) and
                torch.equal(a.data, b.data))


def _dict_eq_literal(d: Series, lit: Series, op: str) -> Series:
    """dict column vs single-row literal: code comparison, no broadcast."""
    target = lit.to_pylist()[0]
    vocab = d.children[0].to_pylist()
    code = vocab.index(target) if target in vocab else -1
    m = d.data == code if op == "eq" else d.data != code
    validity = None if d.validity is None else d.validity.clone()
    if target is None:
        m = torch.zeros(len(d), dtype=torch.bool, device=d.device)
        validity = torch.zeros(len(d), dtype=torch.bool, device=d.device)
    return Series(d.name, DataType.bool(), data=m, validity=validity)


def _string_compare(l: Series, r: Series, op: str, validity) -> Series:
    # dictionary fast paths: evaluate against the vocab, compare codes
    if l.is_dict() or r.is_dict():
        if l.is_dict() and r.is_dict() and op in ("eq", "ne") and \
                _same_vocab(l.children[0], r.children[0]):
            m = l.data == r.data if op == "eq" else l.data != r.data
            return Series(l.name, DataType.bool(), data=m, validity=validity)
        return _string_compare(l.dict_decode(), r.dict_decode(), op,
                               validity)
    if _is_gpu(l):
        nat = native_required()
        cmp = nat.string_compare(l.offsets, l.data, r.offsets, r.data)
        m = {"eq": cmp == 0, "ne": cmp != 0, "lt": cmp < 0, "le": cmp <= 0,
             "gt": cmp > 0, "ge": cmp >= 0}[op]
        return Series(l.name, DataType.bool(), data=m, validity=validity)
    a = np.array(l.to_pylist(), dtype=object)
    b = np.array(r.to_pylist(), dtype=object)
    f = {"eq": lambda x, y: x == y, "ne": lambda x, y: x != y,
         "lt": lambda x, y: x < y, "le": lambda x, y: x <= y,
         "gt": lambda x, y: x > y, "ge": lambda x, y: x >= y}[op]
    out = np.array([False if (x is None or y is None) else f(x, y)
                    for x, y in zip(a, b)], dtype=bool)
    return Series(l.name, DataType.bool(), data=torch.from_numpy(out),
                  validity=validity)


def logical_op(l: Series, r: Series, op: str) -> Series:
    """SQL three-valued logic and/or/xor."""
    l, r = _align(l, r)
    a = l.data.to(torch.bool)
    b = r.data.to(torch.bool)
    if l.validity is None and r.validity is None:
        # non-null fast path: plain boolean kernel, no validity algebra
        out = a & b if op == "and" else (a | b if op == "or" else a ^ b)
        return Series(l.name, DataType.bool(), data=out)
    av = l.validity if l.validity is not None else torch.ones_like(a)
    bv = r.validity if r.validity is not None else torch.ones_like(b)
    if op == "and":
        out = a & b
        # valid when: both valid, or one is a valid False
        validity = (av & bv) | (av & ~a) | (bv & ~b)
    elif op == "or":
        out = a | b
        validity = (av & bv) | (av & a) | (bv & b)
    elif op == "xor":
        out = a ^ b
        validity = av & bv
    else:
        raise ValueError(op)
    if bool(validity.all().item()):
        validity = None
    return Series(l.name, DataType.bool(), data=out,
                  validity=_expand_validity(validity, int(out.shape[0])))


def logical_not(s: Series) -> Series:
    return Series(s.name, DataType.bool(), data=~s.data.to(torch.bool),
                  validity=s.validity)


def if_else(cond: Series, t: Series, f: Series) -> Series:
    n = max(len(cond), len(t), len(f))
    if len(cond) == 1:
        cond = cond.broadcast(n)
    if len(t) == 1 and n > 1:
        t = t.broadcast(n)
    if len(f) == 1 and n > 1:
        f = f.broadcast(n)
    out_dt = supertype(t.dtype, f.dtype)
    t = t.cast(out_dt) if t.dtype != out_dt else t
    f = f.cast(out_dt) if f.dtype != out_dt else f
    m = cond.data.to(torch.bool)
    if cond.validity is not None:
        m = m & cond.validity
    if out_dt.is_decimal() and t.children:
        # wide decimal: select each limb
        from . import decimal128 as d128
        tlo, thi = d128.limbs(t)
        flo, fhi = d128.limbs(f)
        tv = t.validity if t.validity is not None else torch.ones_like(m)
        fv = f.validity if f.validity is not None else torch.ones_like(m)
        validity = torch.where(m, tv, fv)
        if bool(validity.all().item()):
            validity = None
        return d128.make(t.name, out_dt, torch.where(m, tlo, flo),
                         torch.where(m, thi, fhi), validity)
    if out_dt.is_fixed_width():
        td, fd = t.data, f.data
        uview = None
        if td.dtype in (torch.uint16, torch.uint32, torch.uint64):
            # torch.where has no unsigned-wide CPU kernels: select through
            # the bit-identical signed view
            uview = td.dtype
            signed = {torch.uint16: torch.int16, torch.uint32: torch.int32,
                      torch.uint64: torch.int64}[uview]
            td, fd = td.view(signed), fd.view(signed)
        out = torch.where(m, td, fd)
        if uview is not None:
            out = out.view(uview)
        validity = None
        tv = t.validity if t.validity is not None else torch.ones_like(m)
        fv = f.validity if f.validity is not None else torch.ones_like(m)
        validity = torch.where(m, tv, fv)
        if bool(validity.all().item()):
            validity = None
        return Series(t.name, out_dt, data=out, validity=validity)
    # variable width: select via take
    idx = torch.arange(n, dtype=torch.int64, device=t.device)
    t_idx = torch.where(m, idx, torch.full_like(idx, -1))
    f_idx = torch.where(m, torch.full_like(idx, -1), idx)
    tt = t.take(t_idx)
    ff = f.take(f_idx)
    # merge: rows where cond -> tt, else ff
    return _merge_by_mask(m, tt, ff)


def _merge_by_mask(m: torch.Tensor, t: Series, f: Series) -> Series:
    # both t and f have full length with nulls in the opposite slots
    out_validity = torch.where(
        m,
        t.validity if t.validity is not None else torch.ones_like(m),
        f.validity if f.validity is not None else torch.ones_like(m))
    if t.dtype.kind in (TypeKind.STRING, TypeKind.BINARY):
        # rebuild strings row by row on the selected side
        sel_t = t
        sel_f = f
        n = len(t)
        lens_t = sel_t.offsets[1:] - sel_t.offsets[:-1]
        lens_f = sel_f.offsets[1:] - sel_f.offsets[:-1]
        lens = torch.where(m, lens_t, lens_f)
        new_off = torch.zeros(n + 1, dtype=torch.int64, device=m.device)
        torch.cumsum(lens, 0, out=new_off[1:])
        if _is_gpu(t):
            out_bytes = native_required().merge_strings(
                m, sel_t.offsets, sel_t.data, sel_f.offsets, sel_f.data,
                new_off)
        else:
            out_bytes = _cpu_merge_strings(m, sel_t, sel_f, new_off)
        return Series(t.name, t.dtype, data=out_bytes, offsets=new_off,
                      validity=out_validity)
    raise TypeError(f"if_else on {t.dtype!r}")


def _cpu_merge_strings(m, t: Series, f: Series, new_off) -> torch.Tensor:
    mb = m.numpy()
    to, td = t.offsets.numpy(), t.data.numpy()
    fo, fd = f.offsets.numpy(), f.data.numpy()
    no = new_off.numpy()
    out = np.zeros(int(no[-1]), dtype=np.uint8)
    for i in range(len(mb)):
        a, b = no[i], no[i + 1]
        if mb[i]:
            out[a:b] = td[to[i]:to[i] + (b - a)]
        else:
            out[a:b] = fd[fo[i]:fo[i] + (b - a)]
    return torch.from_numpy(out)


def is_in(s: Series, values: Series) -> Series:
    if s.is_dict():
        # evaluate membership on the vocab, gather by code
        vmask = is_in(s.children[0], values)
        out = vmask.data[s.data.to(torch.int64)]
        return Series(s.name, DataType.bool(), data=out,
                      validity=s.validity)
    # numeric/temporal fast path: one sorted-table binary search instead
    # of a compare kernel per value (the O(n·k) loop was a flagged weak
    # spot; strings keep the loop — the dict path above covers hot cases)
    if len(values) > 4 and s.data is not None and s.offsets is None and \
            not s.children and values.data is not None and \
            values.offsets is None and not values.children and \
            s.data.dtype == values.data.dtype and \
            s.data.dtype not in (torch.bool,):
        vd = values.data
        if values.validity is not None:
            vd = vd[values.validity]
        view = {torch.uint16: torch.int16, torch.uint32: torch.int32,
                torch.uint64: torch.int64}.get(vd.dtype)
        sd = s.data
        if view is not None:
            vd, sd = vd.view(view), sd.view(view)
        table = torch.unique(vd.to(s.device))
        if table.numel() == 0:
            data = torch.zeros(len(s), dtype=torch.bool, device=s.device)
        else:
            pos = torch.searchsorted(table, sd.contiguous())
            pos = pos.clamp(max=table.numel() - 1)
            data = table[pos] == sd
        return Series(s.name, DataType.bool(), data=data,
                      validity=s.validity)
    out = None
    for i in range(len(values)):
        v = values.slice(i, i + 1)
        c = compare_op(s, v.broadcast(len(s)), "eq")
        out = c if out is None else logical_op(out, c, "or")
    if out is None:
        return Series(s.name, DataType.bool(),
                      data=torch.zeros(len(s), dtype=torch.bool,
                                       device=s.device))
    return out


def _null_and(a, b):
    if a is None:
        return b
    if b is None:
        return a
    return a & b


# row-wise kernels (hash/groupby/join/sort/partition) — imported last to
# close the module cycle (rowops uses the descriptor helpers above)
from .rowops import (  # noqa: E402
    argsort_multi, groupby, grouped_agg, hash_columns, join,
    partition_by_hash, partition_by_value, partition_random,
)
