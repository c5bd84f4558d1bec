"""Type system: DataType / Field / Schema.

Mirrors the capability surface of the reference type system
(/root/reference/src/daft-schema/src/dtype.rs:17-150) — primitives, temporal
types, strings/binary, nested types and the ML logical types (Embedding,
Image, Tensor, ...) — with a `to_physical()` storage mapping
(dtype.rs:377-423).  Storage here is torch tensors (HBM3E-resident on GPU),
so the physical mapping targets torch dtypes instead of arrow-rs arrays.
"""
from __future__ import annotations

import enum
from dataclasses import dataclass, field as dc_field
from typing import Dict, List, Optional, Tuple

import torch


class TypeKind(enum.Enum):
    NULL = "null"
    BOOL = "bool"
    INT8 = "int8"
    INT16 = "int16"
    INT32 = "int32"
    INT64 = "int64"
    UINT8 = "uint8"
    UINT16 = "uint16"
    UINT32 = "uint32"
    UINT64 = "uint64"
    FLOAT32 = "float32"
    FLOAT64 = "float64"
    DECIMAL128 = "decimal128"
    STRING = "string"
    BINARY = "binary"
    FIXED_SIZE_BINARY = "fixed_size_binary"
    DATE = "date"
    TIME = "time"
    TIMESTAMP = "timestamp"
    DURATION = "duration"
    INTERVAL = "interval"
    LIST = "list"
    FIXED_SIZE_LIST = "fixed_size_list"
    STRUCT = "struct"
    MAP = "map"
    EMBEDDING = "embedding"
    IMAGE = "image"
    FIXED_SHAPE_IMAGE = "fixed_shape_image"
    TENSOR = "tensor"
    FIXED_SHAPE_TENSOR = "fixed_shape_tensor"
    SPARSE_TENSOR = "sparse_tensor"
    PYTHON = "python"
    FILE = "file"
    EXTENSION = "extension"
    UNKNOWN = "unknown"


_INTEGER_KINDS = {
    TypeKind.INT8, TypeKind.INT16, TypeKind.INT32, TypeKind.INT64,
    TypeKind.UINT8, TypeKind.UINT16, TypeKind.UINT32, TypeKind.UINT64,
}
_FLOAT_KINDS = {TypeKind.FLOAT32, TypeKind.FLOAT64}
_SIGNED_INTS = [TypeKind.INT8, TypeKind.INT16, TypeKind.INT32, TypeKind.INT64]
_UNSIGNED_INTS = [TypeKind.UINT8, TypeKind.UINT16, TypeKind.UINT32, TypeKind.UINT64]


@dataclass(frozen=True)
class DataType:
    kind: TypeKind
    # parametric payloads (only the relevant ones are set per kind)
    precision: int = 0            # decimal
    scale: int = 0                # decimal
    timeunit: str = "us"          # time/timestamp/duration
    timezone: Optional[str] = None
    size: int = 0                 # fixed_size_list / fixed_size_binary / embedding
    inner: Optional["DataType"] = None          # list / embedding / tensor
    fields: Tuple["Field", ...] = ()            # struct
    shape: Tuple[int, ...] = ()                 # fixed_shape_tensor / image
    image_mode: Optional[str] = None            # image

    # ---- constructors -------------------------------------------------
    @staticmethod
    def null() -> "DataType": return DataType(TypeKind.NULL)
    @staticmethod
    def bool() -> "DataType": return DataType(TypeKind.BOOL)
    @staticmethod
    def int8() -> "DataType": return DataType(TypeKind.INT8)
    @staticmethod
    def int16() -> "DataType": return DataType(TypeKind.INT16)
    @staticmethod
    def int32() -> "DataType": return DataType(TypeKind.INT32)
    @staticmethod
    def int64() -> "DataType": return DataType(TypeKind.INT64)
    @staticmethod
    def uint8() -> "DataType": return DataType(TypeKind.UINT8)
    @staticmethod
    def uint16() -> "DataType": return DataType(TypeKind.UINT16)
    @staticmethod
    def uint32() -> "DataType": return DataType(TypeKind.UINT32)
    @staticmethod
    def uint64() -> "DataType": return DataType(TypeKind.UINT64)
    @staticmethod
    def float32() -> "DataType": return DataType(TypeKind.FLOAT32)
    @staticmethod
    def float64() -> "DataType": return DataType(TypeKind.FLOAT64)
    @staticmethod
    def decimal128(precision: int, scale: int) -> "DataType":
        return DataType(TypeKind.DECIMAL128, precision=precision, scale=scale)
    @staticmethod
    def string() -> "DataType": return DataType(TypeKind.STRING)
    @staticmethod
    def binary() -> "DataType": return DataType(TypeKind.BINARY)
    @staticmethod
    def fixed_size_binary(size: int) -> "DataType":
        return DataType(TypeKind.FIXED_SIZE_BINARY, size=size)
    @staticmethod
    def date() -> "DataType": return DataType(TypeKind.DATE)
    @staticmethod
    def time(timeunit: str = "us") -> "DataType":
        return DataType(TypeKind.TIME, timeunit=timeunit)
    @staticmethod
    def timestamp(timeunit: str = "us", timezone: Optional[str] = None) -> "DataType":
        return DataType(TypeKind.TIMESTAMP, timeunit=timeunit, timezone=timezone)
    @staticmethod
    def duration(timeunit: str = "us") -> "DataType":
        return DataType(TypeKind.DURATION, timeunit=timeunit)
    @staticmethod
    def interval() -> "DataType": return DataType(TypeKind.INTERVAL)
    @staticmethod
    def list(inner: "DataType") -> "DataType":
        return DataType(TypeKind.LIST, inner=inner)
    @staticmethod
    def fixed_size_list(inner: "DataType", size: int) -> "DataType":
        return DataType(TypeKind.FIXED_SIZE_LIST, inner=inner, size=size)
    @staticmethod
    def struct(fields: Dict[str, "DataType"]) -> "DataType":
        return DataType(TypeKind.STRUCT,
                        fields=tuple(Field(n, t) for n, t in fields.items()))
    @staticmethod
    def map(key: "DataType", value: "DataType") -> "DataType":
        entries = DataType.struct({"key": key, "value": value})
        return DataType(TypeKind.MAP, inner=DataType.list(entries))
    @staticmethod
    def embedding(inner: "DataType", size: int) -> "DataType":
        return DataType(TypeKind.EMBEDDING, inner=inner, size=size)
    @staticmethod
    def image(mode: Optional[str] = None) -> "DataType":
        return DataType(TypeKind.IMAGE, image_mode=mode)
    @staticmethod
    def fixed_shape_image(mode: str, height: int, width: int) -> "DataType":
        return DataType(TypeKind.FIXED_SHAPE_IMAGE, image_mode=mode,
                        shape=(height, width))
    @staticmethod
    def tensor(inner: "DataType") -> "DataType":
        return DataType(TypeKind.TENSOR, inner=inner)
    @staticmethod
    def fixed_shape_tensor(inner: "DataType", shape: Tuple[int, ...]) -> "DataType":
        return DataType(TypeKind.FIXED_SHAPE_TENSOR, inner=inner,
                        shape=tuple(shape))
    @staticmethod
    def sparse_tensor(inner: "DataType") -> "DataType":
        return DataType(TypeKind.SPARSE_TENSOR, inner=inner)
    @staticmethod
    def python() -> "DataType": return DataType(TypeKind.PYTHON)
    @staticmethod
    def file() -> "DataType": return DataType(TypeKind.FILE)

    # ---- predicates ---------------------------------------------------
    def is_null(self) -> bool: return self.kind == TypeKind.NULL
    def is_boolean(self) -> bool: return self.kind == TypeKind.BOOL
    def is_integer(self) -> bool: return self.kind in _INTEGER_KINDS
    def is_signed_integer(self) -> bool: return self.kind in _SIGNED_INTS
    def is_unsigned_integer(self) -> bool: return self.kind in _UNSIGNED_INTS
    def is_floating(self) -> bool: return self.kind in _FLOAT_KINDS
    def is_decimal(self) -> bool: return self.kind == TypeKind.DECIMAL128
    def is_numeric(self) -> bool:
        return self.is_integer() or self.is_floating() or self.is_decimal()
    def is_string(self) -> bool: return self.kind == TypeKind.STRING
    def is_binary(self) -> bool:
        return self.kind in (TypeKind.BINARY, TypeKind.FIXED_SIZE_BINARY)
    def is_temporal(self) -> bool:
        return self.kind in (TypeKind.DATE, TypeKind.TIME, TypeKind.TIMESTAMP,
                             TypeKind.DURATION, TypeKind.INTERVAL)
    def is_nested(self) -> bool:
        return self.kind in (TypeKind.LIST, TypeKind.FIXED_SIZE_LIST,
                             TypeKind.STRUCT, TypeKind.MAP)
    def is_list(self) -> bool:
        return self.kind in (TypeKind.LIST, TypeKind.FIXED_SIZE_LIST)
    def is_python(self) -> bool: return self.kind == TypeKind.PYTHON
    def is_comparable(self) -> bool:
        return (self.is_numeric() or self.is_string() or self.is_temporal()
                or self.is_boolean() or self.is_binary())

    # ---- physical storage mapping (ref: dtype.rs:377-423) -------------
    def to_physical(self) -> "DataType":
        k = self.kind
        if k == TypeKind.DATE:
            return DataType.int32()
        if k in (TypeKind.TIME, TypeKind.TIMESTAMP, TypeKind.DURATION):
            return DataType.int64()
        if k == TypeKind.DECIMAL128:
            # exact storage: scaled int64 for precision <= 18 (value *
            # 10^scale as a 64-bit integer — sums/compares/joins are exact;
            # ref semantics: daft-core Decimal128Array).  Wider decimals
            # (p > 18) store the scaled i128 as TWO int64 limb children
            # (kernels/decimal128.py): exact add/sub/mul/compare/sort/sum
            # via carry arithmetic, identical torch code on CPU and GPU.
            if self.precision <= 18:
                return DataType.int64()
            return DataType.struct({"lo": DataType.int64(),
                                    "hi": DataType.int64()})
        if k == TypeKind.EMBEDDING:
            return DataType.fixed_size_list(self.inner, self.size)
        if k == TypeKind.FIXED_SHAPE_TENSOR:
            n = 1
            for s in self.shape:
                n *= s
            return DataType.fixed_size_list(self.inner, n)
        if k == TypeKind.FIXED_SHAPE_IMAGE:
            mode = self.image_mode or "RGB"
            ch = {"L": 1, "LA": 2, "RGB": 3, "RGBA": 4}.get(mode, 3)
            return DataType.fixed_size_list(
                DataType.uint8(), self.shape[0] * self.shape[1] * ch)
        if k == TypeKind.IMAGE:
            return DataType.struct({
                "data": DataType.binary(),
                "channel": DataType.uint16(),
                "height": DataType.uint32(),
                "width": DataType.uint32(),
                "mode": DataType.uint8(),
            })
        if k == TypeKind.TENSOR:
            return DataType.struct({
                "data": DataType.list(self.inner),
                "shape": DataType.list(DataType.uint64()),
            })
        if k == TypeKind.MAP:
            return self.inner  # list<struct<key,value>>
        return self

    # torch storage dtype for fixed-width physical types
    def to_torch(self) -> torch.dtype:
        phys = self.to_physical()
        m = {
            TypeKind.BOOL: torch.bool,
            TypeKind.INT8: torch.int8,
            TypeKind.INT16: torch.int16,
            TypeKind.INT32: torch.int32,
            TypeKind.INT64: torch.int64,
            TypeKind.UINT8: torch.uint8,
            TypeKind.UINT16: torch.uint16,
            TypeKind.UINT32: torch.uint32,
            TypeKind.UINT64: torch.uint64,
            TypeKind.FLOAT32: torch.float32,
            TypeKind.FLOAT64: torch.float64,
        }
        if phys.kind in m:
            return m[phys.kind]
        raise TypeError(f"{self} has no fixed-width torch storage")

    def is_fixed_width(self) -> bool:
        try:
            self.to_torch()
            return True
        except TypeError:
            return False

    def __repr__(self) -> str:
        k = self.kind
        if k == TypeKind.DECIMAL128:
            return f"Decimal128({self.precision},{self.scale})"
        if k == TypeKind.TIMESTAMP:
            return f"Timestamp({self.timeunit},{self.timezone})"
        if k in (TypeKind.TIME, TypeKind.DURATION):
            return f"{k.value.capitalize()}({self.timeunit})"
        if k == TypeKind.LIST:
            return f"List[{self.inner!r}]"
        if k == TypeKind.FIXED_SIZE_LIST:
            return f"FixedSizeList[{self.inner!r};{self.size}]"
        if k == TypeKind.STRUCT:
            inner = ", ".join(f"{f.name}: {f.dtype!r}" for f in self.fields)
            return f"Struct[{inner}]"
        if k == TypeKind.EMBEDDING:
            return f"Embedding[{self.inner!r};{self.size}]"
        if k == TypeKind.FIXED_SHAPE_TENSOR:
            return f"Tensor[{self.inner!r};{'x'.join(map(str, self.shape))}]"
        return k.value.capitalize() if k != TypeKind.STRING else "Utf8"

    def short_name(self) -> str:
        return self.kind.value


def from_torch_dtype(dt: torch.dtype) -> DataType:
    m = {
        torch.bool: DataType.bool(),
        torch.int8: DataType.int8(),
        torch.int16: DataType.int16(),
        torch.int32: DataType.int32(),
        torch.int64: DataType.int64(),
        torch.uint8: DataType.uint8(),
        torch.uint16: DataType.uint16(),
        torch.uint32: DataType.uint32(),
        torch.uint64: DataType.uint64(),
        torch.float16: DataType.float32(),
        torch.bfloat16: DataType.float32(),
        torch.float32: DataType.float32(),
        torch.float64: DataType.float64(),
    }
    if dt in m:
        return m[dt]
    raise TypeError(f"unsupported torch dtype {dt}")


# ---- numeric type promotion (binary-op supertype) ----------------------

_KIND_ORDER = [
    TypeKind.INT8, TypeKind.INT16, TypeKind.INT32, TypeKind.INT64,
    TypeKind.UINT8, TypeKind.UINT16, TypeKind.UINT32, TypeKind.UINT64,
    TypeKind.FLOAT32, TypeKind.FLOAT64,
]


def _int_width(k: TypeKind) -> int:
    return {"int8": 8, "int16": 16, "int32": 32, "int64": 64,
            "uint8": 8, "uint16": 16, "uint32": 32, "uint64": 64}[k.value]


def decimal_binary_result(a: "DataType", b: "DataType", op: str,
                          a_digits=None, b_digits=None):
    """Result dtype of ARITHMETIC between decimal/integer operands —
    the single rule both static typing (BinaryOp.to_field) and the
    runtime kernel (_decimal_binary) follow, so schema and data dtypes
    never diverge.  Returns None when plain float semantics apply.
    a_digits/b_digits: known digit count of an integer-literal operand
    (defaults to int64's 19)."""
    if not (a.is_decimal() or b.is_decimal()):
        return None
    if op in ("div", "pow", "floordiv", "mod"):
        return DataType.float64()
    if a.is_floating() or b.is_floating():
        return DataType.float64()
    for dt in (a, b):
        if not (dt.is_decimal() or dt.is_integer() or dt.is_null()):
            return None

    def parts(dt, digs):
        if dt.is_decimal():
            return dt.precision, dt.scale
        return (digs if digs is not None else 19), 0

    ap, asc = parts(a, a_digits)
    bp, bsc = parts(b, b_digits)
    if op in ("add", "sub"):
        sc = max(asc, bsc)
        if sc > 18:
            return DataType.float64()
        p = max(ap - asc, bp - bsc) + sc + 1
        if p > 18:
            return DataType.decimal128(max(19, min(38, p)), sc)
        return DataType.decimal128(p, sc)
    if op == "mul":
        sc = asc + bsc
        if sc > 18:
            return DataType.float64()
        if (ap - asc) + (bp - bsc) + sc > 18:
            if sc <= 38 and ap + bp + 1 <= 38:
                return DataType.decimal128(max(19, min(38, ap + bp + 1)),
                                           sc)
            return DataType.float64()
        return DataType.decimal128(min(18, ap + bp), sc)
    return None


def supertype(a: DataType, b: DataType) -> DataType:
    """Least common supertype for binary operations."""
    if a == b:
        return a
    if a.is_null():
        return b
    if b.is_null():
        return a
    if a.is_decimal() or b.is_decimal():
        if a.is_decimal() and b.is_decimal():
            sc = max(a.scale, b.scale)
            ip = max(a.precision - a.scale, b.precision - b.scale)
            p = ip + sc + 1
            if p <= 38:
                return DataType.decimal128(p, sc)
            return DataType.float64()
        dec, other = (a, b) if a.is_decimal() else (b, a)
        if other.is_integer():
            p = min(38, max(dec.precision, 19 + dec.scale))
            return DataType.decimal128(p, dec.scale)
        return DataType.float64()
    if a.is_temporal() or b.is_temporal():
        if a.kind == b.kind:
            return a if a.timeunit >= b.timeunit else b
        if a.is_temporal() and b.is_numeric():
            return a
        if b.is_temporal() and a.is_numeric():
            return b
        return a
    if a.is_string() and b.is_string():
        return a
    if a.is_boolean() and b.is_numeric():
        return b
    if b.is_boolean() and a.is_numeric():
        return a
    if not (a.is_numeric() and b.is_numeric()):
        raise TypeError(f"no supertype for {a} and {b}")
    if a.is_floating() or b.is_floating():
        if a.kind == TypeKind.FLOAT64 or b.kind == TypeKind.FLOAT64:
            return DataType.float64()
        # float32 + wide ints -> float64
        other = b if a.is_floating() else a
        if other.is_integer() and _int_width(other.kind) >= 32:
            return DataType.float64()
        return DataType.float32()
    # integer/integer
    aw, bw = _int_width(a.kind), _int_width(b.kind)
    asig, bsig = a.is_signed_integer(), b.is_signed_integer()
    if asig == bsig:
        return a if aw >= bw else b
    # mixed sign: widen to signed of max(width)*2 capped at 64
    w = max(aw if asig else aw * 2, bw if bsig else bw * 2)
    w = min(w, 64)
    return {8: DataType.int8(), 16: DataType.int16(),
            32: DataType.int32(), 64: DataType.int64()}[w]


@dataclass(frozen=True)
class Field:
    name: str
    dtype: DataType

    def __repr__(self) -> str:
        return f"{self.name}#{self.dtype!r}"


class Schema:
    """Ordered name -> Field mapping (ref: daft-schema Schema)."""

    __slots__ = ("_fields", "_index")

    def __init__(self, fields: List[Field]):
        self._fields: List[Field] = list(fields)
        self._index: Dict[str, int] = {}
        for i, f in enumerate(self._fields):
            if f.name in self._index:
                raise ValueError(f"duplicate field name: {f.name}")
            self._index[f.name] = i

    @staticmethod
    def from_dict(d: Dict[str, DataType]) -> "Schema":
        return Schema([Field(n, t) for n, t in d.items()])

    def __len__(self) -> int:
        return len(self._fields)

    def __iter__(self):
        return iter(self._fields)

    def __contains__(self, name: str) -> bool:
        return name in self._index

    def __getitem__(self, name_or_idx):
        if isinstance(name_or_idx, int):
            return self._fields[name_or_idx]
        try:
            return self._fields[self._index[name_or_idx]]
        except KeyError:
            raise KeyError(
                f"column {name_or_idx!r} not found; schema has {self.names()}")

    def index_of(self, name: str) -> int:
        return self._index[name]

    def names(self) -> List[str]:
        return [f.name for f in self._fields]

    def fields(self) -> List[Field]:
        return list(self._fields)

    def select(self, names: List[str]) -> "Schema":
        return Schema([self[n] for n in names])

    def union(self, other: "Schema", prefer_right: bool = False) -> "Schema":
        out = list(self._fields)
        for f in other:
            if f.name in self._index:
                if prefer_right:
                    out[self._index[f.name]] = f
                else:
                    raise ValueError(f"duplicate field in union: {f.name}")
            else:
                out.append(f)
        return Schema(out)

    def __eq__(self, other) -> bool:
        return isinstance(other, Schema) and self._fields == other._fields

    def __repr__(self) -> str:
        inner = ", ".join(repr(f) for f in self._fields)
        return f"Schema({inner})"
