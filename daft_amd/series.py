"""Device-resident columnar Series.

The MI355X-native analog of the reference's Series/DataArray layer
(/root/reference/src/daft-core/src/array/mod.rs:41, series/ops/*): a named,
typed column whose buffers are torch tensors — HBM3E-resident when on GPU,
host tensors on CPU.  Layout is Arrow-compatible so D2H round-trips to
pyarrow are memcpy-shaped:

  fixed-width : data[n] (+ validity[n] bool, True = valid)
  utf8/binary : offsets[n+1] int64 + bytes[m] uint8 (+ validity)
  list        : offsets[n+1] int64 + child Series (+ validity)
  fixed-size-list / embedding / fixed-shape-tensor : child Series of n*size
  struct      : children Series (+ validity)
  python      : host list of objects

Hot ops (filter/take/hash/sort/groupby/join) dispatch through
daft_amd.kernels, which routes GPU tensors to the hand-written HIP/CDNA4
extension and CPU tensors to torch fallbacks.
"""
from __future__ import annotations

import datetime as _dt
from typing import Any, List, Optional, Sequence, Tuple, Union

import numpy as np
import torch

from .schema import DataType, Field, TypeKind, from_torch_dtype, supertype

_EPOCH = _dt.date(1970, 1, 1)


def _null_and(a: Optional[torch.Tensor], b: Optional[torch.Tensor]):
    if a is None:
        return b
    if b is None:
        return a
    return a & b


class Series:
    __slots__ = ("name", "dtype", "data", "validity", "offsets", "children",
                 "pyobjs", "_length")

    def __init__(self, name: str, dtype: DataType,
                 data: Optional[torch.Tensor] = None,
                 validity: Optional[torch.Tensor] = None,
                 offsets: Optional[torch.Tensor] = None,
                 children: Optional[List["Series"]] = None,
                 pyobjs: Optional[list] = None,
                 length: Optional[int] = None):
        self.name = name
        self.dtype = dtype
        self.data = data
        self.validity = validity
        self.offsets = offsets
        self.children = children or []
        self.pyobjs = pyobjs
        if length is not None:
            self._length = length
        elif offsets is not None:
            self._length = int(offsets.shape[0]) - 1
        elif data is not None:
            self._length = int(data.shape[0])
        elif pyobjs is not None:
            self._length = len(pyobjs)
        elif children:
            k = dtype.kind
            if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING):
                self._length = len(self.children[0]) // max(dtype.size, 1)
            elif k == TypeKind.FIXED_SHAPE_TENSOR:
                n = 1
                for s in dtype.shape:
                    n *= s
                self._length = len(self.children[0]) // max(n, 1)
            else:
                self._length = len(self.children[0])
        else:
            self._length = 0

    # ------------------------------------------------------------------
    def __len__(self) -> int:
        return self._length

    @property
    def device(self) -> torch.device:
        for t in (self.data, self.offsets, self.validity):
            if t is not None:
                return t.device
        if self.children:
            return self.children[0].device
        return torch.device("cpu")

    def is_gpu(self) -> bool:
        return self.device.type == "cuda"

    def field(self) -> Field:
        return Field(self.name, self.dtype)

    def rename(self, name: str) -> "Series":
        out = self._shallow_copy()
        out.name = name
        return out

    def _shallow_copy(self) -> "Series":
        return Series(self.name, self.dtype, self.data, self.validity,
                      self.offsets, list(self.children), self.pyobjs,
                      self._length)

    def with_validity(self, validity: Optional[torch.Tensor]) -> "Series":
        out = self._shallow_copy()
        out.validity = validity
        return out

    def null_count(self) -> int:
        if self.validity is None:
            return 0
        return int((~self.validity).sum().item())

    # ------------------------------------------------------------------
    # dictionary-encoded strings: data = int32 codes, children = [vocab].
    # The GPU-native representation for low-cardinality string columns —
    # takes/filters gather 4-byte codes, predicates evaluate on the vocab,
    # groupby keys compare codes (Arrow dictionary array equivalent).
    # ------------------------------------------------------------------
    def is_dict(self) -> bool:
        return (self.dtype.kind in (TypeKind.STRING, TypeKind.BINARY)
                and self.offsets is None and bool(self.children))

    @staticmethod
    def make_dict(name: str, vocab: "Series", codes: torch.Tensor,
                  validity: Optional[torch.Tensor] = None) -> "Series":
        assert vocab.dtype.kind in (TypeKind.STRING, TypeKind.BINARY)
        assert not vocab.is_dict()
        return Series(name, vocab.dtype, data=codes.to(torch.int32),
                      children=[vocab], validity=validity,
                      length=int(codes.shape[0]))

    def dict_decode(self) -> "Series":
        """Materialize a dictionary-encoded column to plain offsets+bytes."""
        if not self.is_dict():
            return self
        out = self.children[0].take(self.data.to(torch.int64))
        out = out.rename(self.name)
        if self.validity is not None:
            v = self.validity if out.validity is None \
                else (out.validity & self.validity)
            out = out.with_validity(v)
        return out

    def to(self, device, non_blocking: bool = False) -> "Series":
        def mv(t):
            return None if t is None else t.to(device,
                                               non_blocking=non_blocking)
        return Series(self.name, self.dtype, mv(self.data), mv(self.validity),
                      mv(self.offsets),
                      [c.to(device, non_blocking) for c in self.children],
                      self.pyobjs, self._length)

    def cpu_pinned(self) -> "Series":
        """D2H straight into pinned host memory (one DMA, no pageable
        staging) — host-staged tables stream back to HBM at full link
        speed because morsel slices of pinned storage stay pinned."""
        def mv(t):
            if t is None:
                return None
            if not t.is_cuda:
                return t.pin_memory() if not t.is_pinned() else t
            out = torch.empty_like(t, device="cpu", pin_memory=True)
            out.copy_(t)
            return out
        return Series(self.name, self.dtype, mv(self.data),
                      mv(self.validity), mv(self.offsets),
                      [c.cpu_pinned() for c in self.children], self.pyobjs,
                      self._length)

    def is_pinned(self) -> bool:
        for t in (self.data, self.validity, self.offsets):
            if t is not None:
                if t.is_cuda or not t.is_pinned():
                    return False
        return all(c.is_pinned() for c in self.children)

    def pinned(self) -> "Series":
        """Copy host buffers into pinned (page-locked) memory so H2D
        transfers run as async DMA on a copy stream (out-of-core morsel
        staging; ref role: the reference's scan-task reader prefetch,
        sources/scan_task_reader.rs)."""
        def pin(t):
            if t is None or t.is_cuda or t.is_pinned():
                return t
            return t.pin_memory()
        return Series(self.name, self.dtype, pin(self.data),
                      pin(self.validity), pin(self.offsets),
                      [c.pinned() for c in self.children], self.pyobjs,
                      self._length)

    # ------------------------------------------------------------------
    # construction
    # ------------------------------------------------------------------
    @staticmethod
    def from_pylist(name: str, values: Sequence[Any],
                    dtype: Optional[DataType] = None,
                    device: Union[str, torch.device] = "cpu") -> "Series":
        if dtype is None:
            dtype = _infer_dtype(values)
        s = _from_pylist_typed(name, list(values), dtype)
        if str(device) != "cpu":
            s = s.to(device)
        return s

    @staticmethod
    def from_numpy(name: str, arr: np.ndarray,
                   dtype: Optional[DataType] = None) -> "Series":
        if arr.dtype == object or arr.dtype.kind in ("U", "S"):
            return Series.from_pylist(name, arr.tolist(), dtype)
        if arr.dtype == np.uint64:
            t = torch.from_numpy(arr.astype(np.int64)).view(torch.uint64)
        else:
            t = torch.from_numpy(np.ascontiguousarray(arr))
        dt = dtype or from_torch_dtype(t.dtype)
        if t.dtype != dt.to_torch():
            t = t.to(dt.to_torch())
        return Series(name, dt, data=t)

    @staticmethod
    def from_torch(name: str, t: torch.Tensor,
                   dtype: Optional[DataType] = None,
                   validity: Optional[torch.Tensor] = None) -> "Series":
        if t.dim() == 2:
            inner = from_torch_dtype(t.dtype)
            child = Series("item", inner, data=t.reshape(-1).contiguous())
            dt = dtype or DataType.embedding(inner, t.shape[1])
            return Series(name, dt, children=[child], validity=validity,
                          length=t.shape[0])
        dt = dtype or from_torch_dtype(t.dtype)
        return Series(name, dt, data=t.contiguous(), validity=validity)

    @staticmethod
    def from_arrow(name: str, arr) -> "Series":
        import pyarrow as pa
        from . import arrow_interop
        if isinstance(arr, pa.ChunkedArray):
            arr = arr.combine_chunks()
        return arrow_interop.from_arrow_array(name, arr)

    @staticmethod
    def null(name: str, dtype: DataType, length: int,
             device="cpu") -> "Series":
        return full_null(name, dtype, length, device)

    # ------------------------------------------------------------------
    # export
    # ------------------------------------------------------------------
    def cpu(self) -> "Series":
        return self.to("cpu")

    def to_pylist(self) -> list:
        s = self.cpu()
        k = self.dtype.kind
        valid = None if s.validity is None else s.validity.numpy()

        def wrap(vals):
            if valid is None:
                return list(vals)
            return [v if ok else None for v, ok in zip(vals, valid)]

        if k == TypeKind.PYTHON:
            return list(s.pyobjs)
        if s.is_dict():
            vocab = s.children[0].to_pylist()
            codes = s.data.to(torch.int64).numpy()
            if valid is None:
                return [vocab[c] for c in codes]
            return [vocab[c] if ok else None
                    for c, ok in zip(codes, valid)]
        if k in (TypeKind.STRING, TypeKind.BINARY):
            off = s.offsets.numpy()
            buf = s.data.numpy().tobytes() if len(s.data) else b""
            out = []
            for i in range(len(s)):
                if valid is not None and not valid[i]:
                    out.append(None)
                    continue
                b = buf[off[i]:off[i + 1]]
                out.append(b.decode("utf-8", "replace")
                           if k == TypeKind.STRING else b)
            return out
        if k == TypeKind.LIST:
            child = s.children[0].to_pylist()
            off = s.offsets.numpy()
            return wrap([child[off[i]:off[i + 1]] for i in range(len(s))])
        if k == TypeKind.MAP:
            child = s.children[0].to_pylist()
            off = s.offsets.numpy()
            return wrap([{e["key"]: e["value"]
                          for e in child[off[i]:off[i + 1]]}
                         for i in range(len(s))])
        if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING):
            child = s.children[0].to_pylist()
            n = self.dtype.size
            return wrap([child[i * n:(i + 1) * n] for i in range(len(s))])
        if k == TypeKind.FIXED_SHAPE_IMAGE:
            ch = {"L": 1, "LA": 2, "RGB": 3, "RGBA": 4}.get(
                self.dtype.image_mode or "RGB", 3)
            n = self.dtype.shape[0] * self.dtype.shape[1] * ch
            child = s.children[0].to_pylist()
            return wrap([child[i * n:(i + 1) * n] for i in range(len(s))])
        if k == TypeKind.IMAGE:
            cols = [c.to_pylist() for c in s.children]
            names = ["data", "channel", "height", "width", "mode"]
            return wrap([dict(zip(names, vals)) for vals in zip(*cols)])
        if k == TypeKind.FIXED_SHAPE_TENSOR:
            n = 1
            for d in self.dtype.shape:
                n *= d
            flat = s.children[0].data.numpy()
            return wrap([flat[i * n:(i + 1) * n].reshape(self.dtype.shape)
                         for i in range(len(s))])
        if k == TypeKind.STRUCT:
            cols = [c.to_pylist() for c in s.children]
            names = [f.name for f in self.dtype.fields]
            return wrap([dict(zip(names, vals)) for vals in zip(*cols)]
                        if cols else [{}] * len(s))
        if k == TypeKind.DATE:
            days = s.data.numpy()
            return wrap([_EPOCH + _dt.timedelta(days=int(d)) for d in days])
        if k == TypeKind.TIMESTAMP:
            us = {"s": 10**6, "ms": 10**3, "us": 1, "ns": 1}[self.dtype.timeunit]
            vals = s.data.numpy()
            out = []
            for v in vals:
                v = int(v)
                if self.dtype.timeunit == "ns":
                    v = v // 1000
                else:
                    v = v * us
                out.append(_dt.datetime(1970, 1, 1) + _dt.timedelta(microseconds=v))
            return wrap(out)
        if k == TypeKind.DURATION:
            mult = {"s": 10**6, "ms": 10**3, "us": 1, "ns": 1}[
                self.dtype.timeunit]
            vals = s.data.numpy()
            out = []
            for v in vals:
                v = int(v)
                us = v // 1000 if self.dtype.timeunit == "ns" else v * mult
                out.append(_dt.timedelta(microseconds=us))
            return wrap(out)
        if k == TypeKind.DECIMAL128 and s.data is not None and \
                s.data.dtype == torch.int64:
            import decimal as _dec
            return wrap([_dec.Decimal(int(v)).scaleb(-self.dtype.scale)
                         for v in s.data.numpy()])
        if k == TypeKind.DECIMAL128 and s.children:
            import decimal as _dec
            from .kernels import decimal128 as d128
            ints = d128.ints_from_tensors(*d128.limbs(s))
            with _dec.localcontext() as _ctx:
                _ctx.prec = 60
                return wrap([_dec.Decimal(v).scaleb(-self.dtype.scale)
                             for v in ints])
        if k == TypeKind.BOOL:
            return wrap([bool(v) for v in s.data.numpy()])
        vals = s.data
        if vals.dtype == torch.uint64:
            vals = vals.view(torch.int64)
            out = [int(v) & 0xFFFFFFFFFFFFFFFF for v in vals.numpy()]
            return wrap(out)
        out = vals.numpy().tolist()
        return wrap(out)

    def to_arrow(self):
        from . import arrow_interop
        return arrow_interop.to_arrow_array(self.cpu())

    def to_numpy(self) -> np.ndarray:
        s = self.cpu()
        if s.data is not None and self.dtype.is_fixed_width() and s.validity is None:
            d = s.data
            if d.dtype == torch.uint64:
                return d.view(torch.int64).numpy().astype(np.uint64)
            return d.numpy()
        return np.array(self.to_pylist(), dtype=object)

    # ------------------------------------------------------------------
    # selection ops (HIP kernels on GPU — ref: daft-core array/ops/{filter,take,concat}.rs)
    # ------------------------------------------------------------------
    def take(self, indices: torch.Tensor,
             has_neg: Optional[bool] = None) -> "Series":
        """Gather rows by index; index -1 produces null."""
        from . import kernels
        return kernels.take(self, indices, has_neg=has_neg)

    def filter(self, mask: "Series") -> "Series":
        from . import kernels
        idx = kernels.compact_indices(mask)
        return self.take(idx)

    def slice(self, start: int, end: int) -> "Series":
        """Contiguous row slice — tensor VIEWS, not gathers (hot for
        out-of-core morsel streaming: slicing a host partition must not
        copy the table)."""
        n = len(self)
        start = max(0, min(start, n))
        end = max(start, min(end, n))
        if start == 0 and end == n:
            return self
        k = self.dtype.kind
        validity = self.validity[start:end] \
            if self.validity is not None else None
        if k == TypeKind.PYTHON:
            return Series(self.name, self.dtype,
                          pyobjs=self.pyobjs[start:end], validity=validity,
                          length=end - start)
        if self.is_dict():
            return Series.make_dict(self.name, self.children[0],
                                    self.data[start:end], validity)
        if k in (TypeKind.STRING, TypeKind.BINARY):
            offs = self.offsets[start:end + 1]
            lo = int(offs[0])
            hi = int(offs[-1])
            return Series(self.name, self.dtype,
                          data=self.data[lo:hi], offsets=offs - lo,
                          validity=validity)
        if k in (TypeKind.LIST, TypeKind.MAP):
            offs = self.offsets[start:end + 1]
            lo = int(offs[0])
            hi = int(offs[-1])
            return Series(self.name, self.dtype, offsets=offs - lo,
                          children=[self.children[0].slice(lo, hi)],
                          validity=validity)
        if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING,
                 TypeKind.FIXED_SHAPE_TENSOR, TypeKind.FIXED_SHAPE_IMAGE):
            if k in (TypeKind.FIXED_SHAPE_TENSOR, TypeKind.FIXED_SHAPE_IMAGE):
                sz = 1
                for d in self.dtype.shape:
                    sz *= d
                if k == TypeKind.FIXED_SHAPE_IMAGE:
                    sz *= {"L": 1, "LA": 2, "RGB": 3, "RGBA": 4}.get(
                        self.dtype.image_mode or "RGB", 3)
            else:
                sz = self.dtype.size
            return Series(self.name, self.dtype,
                          children=[self.children[0].slice(start * sz,
                                                           end * sz)],
                          validity=validity, length=end - start)
        if k in (TypeKind.STRUCT, TypeKind.IMAGE, TypeKind.TENSOR) or \
                (k == TypeKind.DECIMAL128 and self.children):
            # row-aligned children: slice each
            return Series(self.name, self.dtype,
                          children=[c.slice(start, end)
                                    for c in self.children],
                          validity=validity, length=end - start)
        return Series(self.name, self.dtype, data=self.data[start:end],
                      validity=validity)

    def head(self, n: int) -> "Series":
        return self.slice(0, n)

    @staticmethod
    def concat(series: List["Series"]) -> "Series":
        from . import kernels
        return kernels.concat(series)

    def broadcast(self, n: int) -> "Series":
        if len(self) == n:
            return self
        assert len(self) == 1, "can only broadcast length-1 Series"
        idx = torch.zeros(n, dtype=torch.int64, device=self.device)
        return self.take(idx)

    # ------------------------------------------------------------------
    # casts
    # ------------------------------------------------------------------
    def cast(self, dtype: DataType) -> "Series":
        from . import kernels
        return kernels.cast(self, dtype)

    # ------------------------------------------------------------------
    # elementwise (validity-aware; fixed-width via torch on both devices)
    # ------------------------------------------------------------------
    def _binary_numeric(self, other: "Series", op: str) -> "Series":
        from . import kernels
        return kernels.binary_op(self, other, op)

    def __add__(self, o): return self._binary_numeric(o, "add")
    def __sub__(self, o): return self._binary_numeric(o, "sub")
    def __mul__(self, o): return self._binary_numeric(o, "mul")
    def __truediv__(self, o): return self._binary_numeric(o, "div")
    def __mod__(self, o): return self._binary_numeric(o, "mod")
    def __floordiv__(self, o): return self._binary_numeric(o, "floordiv")

    def compare(self, other: "Series", op: str) -> "Series":
        from . import kernels
        return kernels.compare_op(self, other, op)

    def logical(self, other: "Series", op: str) -> "Series":
        from . import kernels
        return kernels.logical_op(self, other, op)

    def logical_not(self) -> "Series":
        from . import kernels
        return kernels.logical_not(self)

    def is_null(self) -> "Series":
        if self.validity is None:
            t = torch.zeros(len(self), dtype=torch.bool, device=self.device)
        else:
            t = ~self.validity
        return Series(self.name, DataType.bool(), data=t)

    def not_null(self) -> "Series":
        if self.validity is None:
            t = torch.ones(len(self), dtype=torch.bool, device=self.device)
        else:
            t = self.validity.clone()
        return Series(self.name, DataType.bool(), data=t)

    def fill_null(self, fill: "Series") -> "Series":
        from . import kernels
        return kernels.if_else(self.not_null(), self, fill)

    def if_else(self, truthy: "Series", falsy: "Series") -> "Series":
        from . import kernels
        return kernels.if_else(self, truthy, falsy)

    def is_in(self, values: "Series") -> "Series":
        from . import kernels
        return kernels.is_in(self, values)

    def between(self, lo: "Series", hi: "Series") -> "Series":
        ge = self.compare(lo, "ge")
        le = self.compare(hi, "le")
        return ge.logical(le, "and")

    # ------------------------------------------------------------------
    # hashing / sorting handles (ref: daft-core kernels/hashing.rs, ops/sort.rs)
    # ------------------------------------------------------------------
    def hash(self, seed: int = 0) -> torch.Tensor:
        from . import kernels
        return kernels.hash_columns([self], seed)

    def argsort(self, descending: bool = False,
                nulls_first: bool = False) -> torch.Tensor:
        from . import kernels
        return kernels.argsort_multi([self], [descending], [nulls_first])

    # ------------------------------------------------------------------
    def __repr__(self) -> str:
        head = self.to_pylist()[:8] if len(self) <= 64 else self.head(8).to_pylist()
        return (f"Series[{self.name}: {self.dtype!r}; n={len(self)}; "
                f"dev={self.device}] {head}")


# ---------------------------------------------------------------------------
# construction helpers
# ---------------------------------------------------------------------------

def _infer_dtype(values: Sequence[Any]) -> DataType:
    for v in values:
        if v is None:
            continue
        if isinstance(v, bool):
            return DataType.bool()
        if isinstance(v, int):
            return DataType.int64()
        if isinstance(v, float):
            return DataType.float64()
        if type(v).__name__ == "Decimal":
            # precision/scale must cover EVERY value, not just the first
            # (a leading small value would infer p<=18 and overflow the
            # scaled-int64 storage on a later wide one)
            scl = 0
            digs = 1
            for w in values:
                if w is None or type(w).__name__ != "Decimal":
                    continue
                t = w.as_tuple()
                s_ = max(0, -t.exponent) if isinstance(t.exponent, int) \
                    else 0
                scl = max(scl, s_)
                digs = max(digs, len(t.digits) + max(0, s_ - (-t.exponent
                           if isinstance(t.exponent, int) else 0)))
            # integer digits + final scale
            idigs = 0
            for w in values:
                if w is None or type(w).__name__ != "Decimal":
                    continue
                t = w.as_tuple()
                exp = t.exponent if isinstance(t.exponent, int) else 0
                idigs = max(idigs, len(t.digits) + exp)
            digs = max(digs, idigs + scl, scl + 1)
            return DataType.decimal128(min(digs, 38), scl)
        if isinstance(v, str):
            return DataType.string()
        if isinstance(v, bytes):
            return DataType.binary()
        if isinstance(v, _dt.datetime):
            return DataType.timestamp("us")
        if isinstance(v, _dt.timedelta):
            return DataType.duration("us")
        if isinstance(v, _dt.date):
            return DataType.date()
        if isinstance(v, (list, tuple)):
            return DataType.list(_infer_dtype(v))
        if isinstance(v, np.ndarray):
            return DataType.tensor(from_torch_dtype(
                torch.from_numpy(v[:0].copy()).dtype))
        if isinstance(v, dict):
            return DataType.struct({k: _infer_dtype([vv]) for k, vv in v.items()})
        if isinstance(v, (np.integer,)):
            return DataType.int64()
        if isinstance(v, (np.floating,)):
            return DataType.float64()
        return DataType.python()
    return DataType.null()


def _validity_from(values: list, device="cpu") -> Optional[torch.Tensor]:
    if any(v is None for v in values):
        return torch.tensor([v is not None for v in values], dtype=torch.bool)
    return None


def _from_pylist_typed(name: str, values: list, dtype: DataType) -> Series:
    k = dtype.kind
    n = len(values)
    validity = _validity_from(values)
    if k == TypeKind.NULL:
        return full_null(name, DataType.null(), n)
    if k == TypeKind.PYTHON:
        return Series(name, dtype, pyobjs=values,
                      validity=validity, length=n)
    if k in (TypeKind.STRING, TypeKind.BINARY):
        bufs = []
        offs = np.zeros(n + 1, dtype=np.int64)
        pos = 0
        for i, v in enumerate(values):
            if v is None:
                offs[i + 1] = pos
                continue
            b = v.encode("utf-8") if isinstance(v, str) else bytes(v)
            bufs.append(b)
            pos += len(b)
            offs[i + 1] = pos
        blob = b"".join(bufs)
        data = torch.frombuffer(bytearray(blob), dtype=torch.uint8) \
            if blob else torch.zeros(0, dtype=torch.uint8)
        return Series(name, dtype, data=data, validity=validity,
                      offsets=torch.from_numpy(offs))
    if k == TypeKind.DATE:
        days = [0 if v is None else (v - _EPOCH).days if isinstance(v, _dt.date)
                else int(v) for v in values]
        return Series(name, dtype,
                      data=torch.tensor(days, dtype=torch.int32),
                      validity=validity)
    if k == TypeKind.TIMESTAMP:
        mult = {"s": 1, "ms": 10**3, "us": 10**6, "ns": 10**9}[dtype.timeunit]
        out = []
        for v in values:
            if v is None:
                out.append(0)
            elif isinstance(v, _dt.datetime):
                out.append(int(v.timestamp() * mult) if v.tzinfo else
                           int((v - _dt.datetime(1970, 1, 1)).total_seconds() * mult))
            else:
                out.append(int(v))
        return Series(name, dtype, data=torch.tensor(out, dtype=torch.int64),
                      validity=validity)
    if k == TypeKind.DURATION:
        mult = {"s": 1, "ms": 10**3, "us": 10**6, "ns": 10**9}[dtype.timeunit]
        out = []
        for v in values:
            if v is None:
                out.append(0)
            elif isinstance(v, _dt.timedelta):
                out.append(int(v.total_seconds() * mult))
            else:
                out.append(int(v))
        return Series(name, dtype, data=torch.tensor(out, dtype=torch.int64),
                      validity=validity)
    if k == TypeKind.LIST:
        offs = np.zeros(n + 1, dtype=np.int64)
        flat = []
        for i, v in enumerate(values):
            if v is not None:
                flat.extend(v)
            offs[i + 1] = len(flat)
        child = _from_pylist_typed("item", flat, dtype.inner)
        return Series(name, dtype, offsets=torch.from_numpy(offs),
                      children=[child], validity=validity)
    if k == TypeKind.MAP:
        # logical Map over list<struct<key, value>> physical storage
        # (ref: daft-schema dtype.rs Map -> List(Struct) to_physical)
        entries = []
        for v in values:
            if v is None:
                entries.append(None)
            elif isinstance(v, dict):
                entries.append([{"key": kk, "value": vv}
                                for kk, vv in v.items()])
            else:               # already [(k, v)] pairs or entry structs
                entries.append([e if isinstance(e, dict) and
                                "key" in e else
                                {"key": e[0], "value": e[1]}
                                for e in v])
        phys = _from_pylist_typed(name, entries, dtype.inner)
        return Series(name, dtype, offsets=phys.offsets,
                      children=phys.children, validity=validity)
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING):
        sz = dtype.size
        flat = []
        for v in values:
            if v is None:
                flat.extend([None] * sz)
            else:
                assert len(v) == sz, f"expected list of size {sz}"
                flat.extend(v)
        child = _from_pylist_typed("item", flat, dtype.inner)
        return Series(name, dtype, children=[child], validity=validity,
                      length=n)
    if k == TypeKind.FIXED_SHAPE_TENSOR:
        sz = 1
        for d in dtype.shape:
            sz *= d
        flat = []
        for v in values:
            if v is None:
                flat.extend([0] * sz)
            else:
                arr = np.asarray(v).reshape(-1)
                assert arr.size == sz
                flat.extend(arr.tolist())
        child = _from_pylist_typed("item", flat, dtype.inner)
        return Series(name, dtype, children=[child], validity=validity,
                      length=n)
    if k == TypeKind.STRUCT:
        children = []
        for f in dtype.fields:
            vals = [None if v is None else v.get(f.name) for v in values]
            children.append(_from_pylist_typed(f.name, vals, f.dtype))
        return Series(name, dtype, children=children, validity=validity,
                      length=n)
    if k == TypeKind.DECIMAL128:
        import decimal as _dec
        with _dec.localcontext() as _ctx:
            _ctx.prec = 60          # default 28 rounds p>28 digit values
            scl = _dec.Decimal(1).scaleb(dtype.scale)
            ints = [0 if v is None else
                    int((_dec.Decimal(str(v)) * scl).to_integral_value(
                        rounding=_dec.ROUND_HALF_EVEN)) for v in values]
        if dtype.to_physical().kind == TypeKind.INT64:
            return Series(name, dtype,
                          data=torch.tensor(ints, dtype=torch.int64),
                          validity=validity)
        # wide (p>18): scaled i128 as two int64 limbs
        from .kernels import decimal128 as d128
        lo, hi = d128.tensors_from_ints(ints)
        return d128.make(name, dtype, lo, hi, validity)
        vals = [0.0 if v is None else float(v) for v in values]
        return Series(name, dtype,
                      data=torch.tensor(vals, dtype=torch.float64),
                      validity=validity)
    # fixed-width primitives
    tdt = dtype.to_torch()
    if tdt == torch.bool:
        vals = [False if v is None else bool(v) for v in values]
    elif tdt.is_floating_point:
        vals = [0.0 if v is None else float(v) for v in values]
    else:
        vals = [0 if v is None else int(v) for v in values]
    if tdt in (torch.uint16, torch.uint32, torch.uint64):
        base = {torch.uint16: torch.int16, torch.uint32: torch.int32,
                torch.uint64: torch.int64}[tdt]
        mask = (1 << {torch.int16: 16, torch.int32: 32, torch.int64: 64}[base]) - 1
        t = torch.tensor([((v & mask) - (mask + 1) if v > mask // 2 else v)
                          for v in vals], dtype=base).view(tdt)
    else:
        t = torch.tensor(vals, dtype=tdt)
    return Series(name, dtype, data=t, validity=validity)


def full_null(name: str, dtype: DataType, length: int, device="cpu") -> Series:
    validity = torch.zeros(length, dtype=torch.bool, device=device)
    k = dtype.kind
    if k == TypeKind.NULL:
        return Series(name, dtype,
                      data=torch.zeros(length, dtype=torch.bool, device=device),
                      validity=validity)
    if k == TypeKind.PYTHON:
        return Series(name, dtype, pyobjs=[None] * length, validity=validity,
                      length=length)
    if k in (TypeKind.STRING, TypeKind.BINARY):
        return Series(name, dtype,
                      data=torch.zeros(0, dtype=torch.uint8, device=device),
                      offsets=torch.zeros(length + 1, dtype=torch.int64,
                                          device=device),
                      validity=validity)
    if k == TypeKind.LIST:
        child = empty_series("item", dtype.inner, device)
        return Series(name, dtype, offsets=torch.zeros(
            length + 1, dtype=torch.int64, device=device),
            children=[child], validity=validity)
    if k == TypeKind.MAP:
        # child is the entries struct (Map stores list<struct> physically)
        child = empty_series("entries", dtype.inner.inner, device)
        return Series(name, dtype, offsets=torch.zeros(
            length + 1, dtype=torch.int64, device=device),
            children=[child], validity=validity)
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING):
        child = full_null("item", dtype.inner, length * dtype.size, device)
        return Series(name, dtype, children=[child], validity=validity,
                      length=length)
    if k == TypeKind.STRUCT:
        children = [full_null(f.name, f.dtype, length, device)
                    for f in dtype.fields]
        return Series(name, dtype, children=children, validity=validity,
                      length=length)
    if k == TypeKind.DECIMAL128 and dtype.precision > 18:
        z = torch.zeros(length, dtype=torch.int64, device=device)
        return Series(name, dtype,
                      children=[Series("lo", DataType.int64(), data=z),
                                Series("hi", DataType.int64(),
                                       data=z.clone())],
                      validity=validity, length=length)
    t = torch.zeros(length, dtype=dtype.to_torch(), device=device)
    return Series(name, dtype, data=t, validity=validity)


def empty_series(name: str, dtype: DataType, device="cpu") -> Series:
    return full_null(name, dtype, 0, device)


def lit_series(name: str, value: Any, dtype: Optional[DataType] = None,
               device="cpu") -> Series:
    dt = dtype or _infer_dtype([value])
    return Series.from_pylist(name, [value], dt, device=device)
