"""Runtime loader for native extension plugins (capability of the
reference's daft.load_extension + daft-ext C ABI,
/root/reference/src/daft-ext/src/abi/ and daft/runners/flotilla.py:102-137
DAFT_EXTENSION_PATHS propagation).

A plugin is a `.so` built against `daft_amd/ext/daft_ext.h`.  Columns
cross the boundary as raw buffer pointers (torch tensor storage), so a
hipcc-built plugin can launch gfx950 kernels on HBM-resident columns
directly.  See examples/ext_plugin/ for a worked example."""
from __future__ import annotations

import ctypes
import os
from typing import Dict, List, Tuple

import torch

from ..schema import DataType, TypeKind
from ..series import Series

ABI_V1 = 1

_DTYPE_CODE = {
    "int64": 0, "float64": 1, "float32": 2, "int32": 3, "uint8": 4,
    "bool": 5,
}
_CODE_TORCH = {0: torch.int64, 1: torch.float64, 2: torch.float32,
               3: torch.int32, 4: torch.uint8, 5: torch.bool}
_CODE_DTYPE = {0: DataType.int64, 1: DataType.float64, 2: DataType.float32,
               3: DataType.int32, 4: DataType.uint8, 5: DataType.bool}


class _Column(ctypes.Structure):
    _fields_ = [("data", ctypes.c_void_p),
                ("length", ctypes.c_int64),
                ("dtype", ctypes.c_int32),
                ("device", ctypes.c_int32),
                ("validity", ctypes.POINTER(ctypes.c_uint8))]


_SCALAR_FN = ctypes.CFUNCTYPE(
    ctypes.c_int, ctypes.POINTER(_Column), ctypes.c_int32,
    ctypes.POINTER(_Column), ctypes.c_char_p, ctypes.c_int32)

_REGISTER_CB = ctypes.CFUNCTYPE(
    ctypes.c_int, ctypes.c_void_p, ctypes.c_char_p, ctypes.c_int32,
    _SCALAR_FN)


class _Api(ctypes.Structure):
    _fields_ = [("abi_version", ctypes.c_int32),
                ("host", ctypes.c_void_p),
                ("register_scalar", _REGISTER_CB)]


class ExtensionRegistry:
    def __init__(self):
        # name -> (out_dtype_code, fn_ptr); keep libs + callbacks alive
        self.fns: Dict[str, Tuple[int, object]] = {}
        self._libs: List[object] = []
        self._keepalive: List[object] = []


_REGISTRY = ExtensionRegistry()


def load_extension(path: str) -> List[str]:
    """dlopen a plugin and run its registration; returns the names of
    functions it registered."""
    if not os.path.exists(path):
        raise FileNotFoundError(path)
    lib = ctypes.CDLL(path, mode=ctypes.RTLD_LOCAL)
    ver = lib.daft_ext_abi_version()
    if ver != ABI_V1:
        raise RuntimeError(f"extension {path!r} has ABI version {ver}, "
                           f"host supports {ABI_V1}")
    added: List[str] = []

    @_REGISTER_CB
    def register_scalar(_host, name, out_dtype, fn):
        nm = name.decode()
        _REGISTRY.fns[nm] = (int(out_dtype), fn)
        _REGISTRY._keepalive.append(fn)
        added.append(nm)
        return 0

    api = _Api(ABI_V1, None, register_scalar)
    rc = lib.daft_ext_register(ctypes.byref(api))
    if rc != 0:
        raise RuntimeError(f"extension {path!r} registration failed "
                           f"(rc={rc})")
    _REGISTRY._libs.append(lib)
    _REGISTRY._keepalive.append(register_scalar)
    return added


def _series_to_column(s: Series) -> Tuple[_Column, List[object]]:
    code = _DTYPE_CODE.get(s.dtype.to_physical().kind.value)
    if code is None:
        raise TypeError(f"extension functions take fixed-width numeric "
                        f"columns; got {s.dtype!r}")
    keep: List[object] = []
    d = s.data.contiguous()
    keep.append(d)
    col = _Column()
    col.data = ctypes.c_void_p(d.data_ptr())
    col.length = len(s)
    col.dtype = code
    col.device = 1 if d.is_cuda else 0
    if s.validity is not None:
        v = s.validity.to(torch.uint8).contiguous()
        keep.append(v)
        col.validity = ctypes.cast(ctypes.c_void_p(v.data_ptr()),
                                   ctypes.POINTER(ctypes.c_uint8))
    else:
        col.validity = None
    return col, keep


def call_extension_fn(name: str, args: List[Series]) -> Series:
    if name not in _REGISTRY.fns:
        raise KeyError(f"no extension function {name!r} loaded "
                       f"(loaded: {sorted(_REGISTRY.fns)})")
    out_code, fn = _REGISTRY.fns[name]
    n = len(args[0]) if args else 0
    dev = args[0].device if args else "cpu"
    keep: List[object] = []
    carr = (_Column * max(1, len(args)))()
    for i, s in enumerate(args):
        carr[i], k = _series_to_column(s)
        keep.extend(k)
    out_t = torch.empty(n, dtype=_CODE_TORCH[out_code],
                        device=args[0].data.device if args else "cpu")
    out_v = torch.ones(n, dtype=torch.uint8,
                       device=args[0].data.device if args else "cpu")
    out = _Column()
    out.data = ctypes.c_void_p(out_t.data_ptr())
    out.length = n
    out.dtype = out_code
    out.device = 1 if out_t.is_cuda else 0
    out.validity = ctypes.cast(ctypes.c_void_p(out_v.data_ptr()),
                               ctypes.POINTER(ctypes.c_uint8))
    err = ctypes.create_string_buffer(512)
    rc = fn(carr, len(args), ctypes.byref(out), err, 512)
    if rc != 0:
        raise RuntimeError(f"extension function {name!r} failed: "
                           f"{err.value.decode(errors='replace')}")
    if out_t.is_cuda:
        torch.cuda.synchronize()
    validity = None
    if not bool(out_v.all()):
        validity = out_v.to(torch.bool)
    dt = _CODE_DTYPE[out_code]()
    res = Series(name, dt, data=out_t if dt.kind != TypeKind.BOOL
                 else out_t.to(torch.bool), validity=validity)
    return res


def ext_function(name: str, *args):
    """Build an expression calling a loaded extension function."""
    from ..expressions.expressions import Expression, ScalarFn, _to_node
    if name not in _REGISTRY.fns:
        raise KeyError(f"no extension function {name!r} loaded")
    out_code, _ = _REGISTRY.fns[name]
    dt = _CODE_DTYPE[out_code]()

    def run(*series):
        return call_extension_fn(name, list(series))
    nodes = [_to_node(a) for a in args]
    return Expression(ScalarFn(name, run, nodes, dt))
