"""pyarrow <-> Series conversion (exact Arrow physical layouts, so transfers
are memcpy-shaped; ref capability: daft-recordbatch/src/ffi.rs,
common/arrow-ffi).  Host-side only — GPU columns round-trip through .cpu().
"""
from __future__ import annotations

import numpy as np
import torch

from .schema import DataType, TypeKind
from .series import Series


def _pa():
    import pyarrow as pa
    return pa


def dtype_to_arrow(dt: DataType):
    pa = _pa()
    k = dt.kind
    m = {
        TypeKind.NULL: pa.null(), TypeKind.BOOL: pa.bool_(),
        TypeKind.INT8: pa.int8(), TypeKind.INT16: pa.int16(),
        TypeKind.INT32: pa.int32(), TypeKind.INT64: pa.int64(),
        TypeKind.UINT8: pa.uint8(), TypeKind.UINT16: pa.uint16(),
        TypeKind.UINT32: pa.uint32(), TypeKind.UINT64: pa.uint64(),
        TypeKind.FLOAT32: pa.float32(), TypeKind.FLOAT64: pa.float64(),
        TypeKind.STRING: pa.large_string(), TypeKind.BINARY: pa.large_binary(),
        TypeKind.DATE: pa.date32(),
    }
    if k in m:
        return m[k]
    if k == TypeKind.TIMESTAMP:
        return pa.timestamp(dt.timeunit, dt.timezone)
    if k == TypeKind.DURATION:
        return pa.duration(dt.timeunit)
    if k == TypeKind.TIME:
        return pa.time64(dt.timeunit if dt.timeunit in ("us", "ns") else "us")
    if k == TypeKind.DECIMAL128:
        return pa.decimal128(dt.precision, dt.scale)
    if k == TypeKind.LIST:
        return pa.large_list(dtype_to_arrow(dt.inner))
    if k == TypeKind.MAP:
        entries = dt.inner.inner
        return pa.map_(dtype_to_arrow(entries.fields[0].dtype),
                       dtype_to_arrow(entries.fields[1].dtype))
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING):
        return pa.list_(dtype_to_arrow(dt.inner), dt.size)
    if k == TypeKind.FIXED_SHAPE_TENSOR:
        n = 1
        for s in dt.shape:
            n *= s
        return pa.list_(dtype_to_arrow(dt.inner), n)
    if k == TypeKind.STRUCT:
        return pa.struct([(f.name, dtype_to_arrow(f.dtype))
                          for f in dt.fields])
    raise TypeError(f"no arrow mapping for {dt!r}")


def dtype_from_arrow(t) -> DataType:
    pa = _pa()
    if pa.types.is_null(t):
        return DataType.null()
    if pa.types.is_boolean(t):
        return DataType.bool()
    for name in ("int8", "int16", "int32", "int64", "uint8", "uint16",
                 "uint32", "uint64"):
        if getattr(pa.types, f"is_{name}")(t):
            return getattr(DataType, name)()
    if pa.types.is_float16(t) or pa.types.is_float32(t):
        return DataType.float32()
    if pa.types.is_float64(t):
        return DataType.float64()
    if pa.types.is_decimal(t):
        return DataType.decimal128(t.precision, t.scale)
    if pa.types.is_string(t) or pa.types.is_large_string(t):
        return DataType.string()
    if pa.types.is_binary(t) or pa.types.is_large_binary(t) or \
            pa.types.is_fixed_size_binary(t):
        return DataType.binary()
    if pa.types.is_date(t):
        return DataType.date()
    if pa.types.is_timestamp(t):
        return DataType.timestamp(t.unit, t.tz)
    if pa.types.is_duration(t):
        return DataType.duration(t.unit)
    if pa.types.is_time(t):
        return DataType.time("us")
    if pa.types.is_map(t):
        return DataType.map(dtype_from_arrow(t.key_type),
                            dtype_from_arrow(t.item_type))
    if pa.types.is_list(t) or pa.types.is_large_list(t):
        return DataType.list(dtype_from_arrow(t.value_type))
    if pa.types.is_fixed_size_list(t):
        return DataType.fixed_size_list(dtype_from_arrow(t.value_type),
                                        t.list_size)
    if pa.types.is_struct(t):
        return DataType.struct({t.field(i).name: dtype_from_arrow(t.field(i).type)
                                for i in range(t.num_fields)})
    if pa.types.is_dictionary(t):
        return dtype_from_arrow(t.value_type)
    raise TypeError(f"unsupported arrow type {t}")


def from_arrow_array(name: str, arr) -> Series:
    pa = _pa()
    if isinstance(arr, pa.ChunkedArray):
        arr = arr.combine_chunks()
    if pa.types.is_dictionary(arr.type):
        # preserve dictionary encoding (codes + vocab)
        import numpy as _np
        vt = dtype_from_arrow(arr.type.value_type)
        if vt.kind in (TypeKind.STRING, TypeKind.BINARY):
            vocab = from_arrow_array("vocab", arr.dictionary)
            codes = torch.from_numpy(
                _np.ascontiguousarray(
                    arr.indices.to_numpy(zero_copy_only=False)
                    .astype(_np.int32)))
            validity = None
            if arr.null_count:
                validity = torch.from_numpy(
                    arr.is_valid().to_numpy(zero_copy_only=False))
            from .series import Series as _S
            return _S.make_dict(name, vocab, codes, validity)
        arr = arr.cast(arr.type.value_type)
    dt = dtype_from_arrow(arr.type)
    k = dt.kind

    validity = None
    if arr.null_count:
        validity = torch.from_numpy(
            arr.is_valid().to_numpy(zero_copy_only=False))

    if k in (TypeKind.STRING, TypeKind.BINARY):
        # normalize to large offsets
        want = pa.large_string() if k == TypeKind.STRING else pa.large_binary()
        if arr.type != want:
            arr = arr.cast(want)
        bufs = arr.buffers()
        off = np.frombuffer(bufs[1], dtype=np.int64,
                            count=len(arr) + 1 + arr.offset)[arr.offset:]
        if arr.offset:
            off = off - off[0]
        nbytes = int(off[-1])
        start = int(np.frombuffer(bufs[1], dtype=np.int64,
                                  count=arr.offset + 1)[arr.offset]) if arr.offset else 0
        data = np.frombuffer(bufs[2], dtype=np.uint8,
                             count=start + nbytes)[start:] if bufs[2] is not None \
            else np.zeros(0, np.uint8)
        return Series(name, dt, data=torch.from_numpy(data.copy()),
                      offsets=torch.from_numpy(np.ascontiguousarray(off).copy()),
                      validity=validity)
    if k == TypeKind.MAP:
        off = np.asarray(arr.offsets).astype(np.int64)
        keys = from_arrow_array("key", arr.keys)
        items = from_arrow_array("value", arr.items)
        entries = Series("entries", dt.inner.inner,
                         children=[keys, items], length=len(keys))
        return Series(name, dt, offsets=torch.from_numpy(off),
                      children=[entries], validity=validity)
    if k == TypeKind.LIST:
        if not pa.types.is_large_list(arr.type):
            arr = arr.cast(pa.large_list(arr.type.value_type))
        off = np.asarray(arr.offsets)
        child = from_arrow_array("item", arr.values)
        return Series(name, dt, offsets=torch.from_numpy(off.astype(np.int64)),
                      children=[child], validity=validity)
    if k == TypeKind.FIXED_SIZE_LIST:
        child = from_arrow_array("item", arr.values)
        return Series(name, dt, children=[child], validity=validity,
                      length=len(arr))
    if k == TypeKind.STRUCT:
        children = [from_arrow_array(dt.fields[i].name, arr.field(i))
                    for i in range(len(dt.fields))]
        return Series(name, dt, children=children, validity=validity,
                      length=len(arr))
    if k == TypeKind.DECIMAL128:
        if dt.to_physical().kind == TypeKind.INT64:
            # exact: arrow decimal128 stores scaled i128 little-endian; for
            # p <= 18 every value fits the low 64 bits (two's complement)
            a2 = arr.combine_chunks() if hasattr(arr, "combine_chunks") \
                else arr
            buf = a2.buffers()[1]
            off = a2.offset
            raw = np.frombuffer(buf, dtype=np.int64,
                                count=2 * (off + len(a2)))
            lows = np.ascontiguousarray(raw[2 * off::2][:len(a2)])
            return Series(name, dt, data=torch.from_numpy(lows.copy()),
                          validity=validity)
        # wide (p > 18): both little-endian limbs -> struct<lo, hi>
        a2 = arr.combine_chunks() if hasattr(arr, "combine_chunks") else arr
        buf = a2.buffers()[1]
        off = a2.offset
        raw = np.frombuffer(buf, dtype=np.int64, count=2 * (off + len(a2)))
        lo = np.ascontiguousarray(raw[2 * off::2][:len(a2)])
        hi = np.ascontiguousarray(raw[2 * off + 1::2][:len(a2)])
        from .kernels import decimal128 as d128
        return d128.make(name, dt, torch.from_numpy(lo.copy()),
                         torch.from_numpy(hi.copy()), validity)
    if k == TypeKind.NULL:
        from .series import full_null
        return full_null(name, DataType.null(), len(arr))
    # fixed width
    np_arr = arr.to_numpy(zero_copy_only=False)
    if np_arr.dtype == object or np_arr.dtype.kind in ("M", "m"):
        # temporal: use raw storage
        storage = arr.cast(pa.int64() if dt.to_physical().kind == TypeKind.INT64
                           else pa.int32())
        np_arr = storage.to_numpy(zero_copy_only=False)
    np_arr = np.ascontiguousarray(np_arr)
    if np_arr.dtype.kind == "f" and validity is not None:
        np_arr = np.nan_to_num(np_arr)
    if np_arr.dtype == np.uint64:
        t = torch.from_numpy(np_arr.view(np.int64).copy()).view(torch.uint64)
    elif np_arr.dtype == np.uint32:
        t = torch.from_numpy(np_arr.view(np.int32).copy()).view(torch.uint32)
    elif np_arr.dtype == np.uint16:
        t = torch.from_numpy(np_arr.view(np.int16).copy()).view(torch.uint16)
    else:
        t = torch.from_numpy(np_arr.copy())
    if t.dtype != dt.to_torch():
        t = t.to(dt.to_torch())
    return Series(name, dt, data=t, validity=validity)


def to_arrow_array(s: Series):
    pa = _pa()
    if s.is_dict():
        vocab = to_arrow_array(s.children[0])
        idx = s.data.numpy()
        mask = None if s.validity is None else ~s.validity.numpy()
        indices = pa.array(idx, mask=mask)
        return pa.DictionaryArray.from_arrays(indices, vocab)
    dt = s.dtype
    k = dt.kind
    atype = dtype_to_arrow(dt) if k != TypeKind.PYTHON else None
    mask = None
    if s.validity is not None:
        mask = ~s.validity.numpy()

    if k == TypeKind.PYTHON:
        raise TypeError("python-object column cannot convert to arrow")
    if k in (TypeKind.STRING, TypeKind.BINARY):
        off = pa.py_buffer(s.offsets.numpy().tobytes())
        data = pa.py_buffer(s.data.numpy().tobytes())
        vbuf = None
        if mask is not None:
            vbuf = pa.array(~mask).buffers()[1]
        return pa.Array.from_buffers(atype, len(s), [vbuf, off, data])
    if k == TypeKind.MAP:
        entries = s.children[0]
        keys = to_arrow_array(entries.children[0])
        items = to_arrow_array(entries.children[1])
        off32 = pa.array(s.offsets.numpy().astype(np.int32))
        m = pa.MapArray.from_arrays(off32, keys, items)
        if mask is not None:
            import pyarrow.compute as _pc
            # rebuild with validity via take-on-null trick
            idx = pa.array([None if bad else i
                            for i, bad in enumerate(mask)],
                           type=pa.int64())
            m = m.take(idx)
        return m
    if k == TypeKind.LIST:
        child = to_arrow_array(s.children[0])
        off = pa.py_buffer(s.offsets.numpy().tobytes())
        vbuf = pa.array(~mask).buffers()[1] if mask is not None else None
        return pa.Array.from_buffers(atype, len(s), [vbuf, off],
                                     children=[child])
    if k in (TypeKind.FIXED_SIZE_LIST, TypeKind.EMBEDDING,
             TypeKind.FIXED_SHAPE_TENSOR):
        child = to_arrow_array(s.children[0])
        vbuf = pa.array(~mask).buffers()[1] if mask is not None else None
        return pa.Array.from_buffers(atype, len(s), [vbuf], children=[child])
    if k == TypeKind.STRUCT:
        children = [to_arrow_array(c) for c in s.children]
        vbuf = pa.array(~mask).buffers()[1] if mask is not None else None
        return pa.Array.from_buffers(atype, len(s), [vbuf], children=children)
    if k == TypeKind.DECIMAL128:
        if s.data is not None and s.data.dtype == torch.int64:
            lows = s.data.numpy()
            i128 = np.empty((len(s), 2), dtype=np.int64)
            i128[:, 0] = lows
            i128[:, 1] = lows >> 63  # sign extension into the high half
            vbuf = pa.array(~mask).buffers()[1] if mask is not None \
                else None
            return pa.Array.from_buffers(
                atype, len(s), [vbuf, pa.py_buffer(i128.tobytes())])
        # wide (p > 18): interleave the limbs into the 16-byte layout
        from .kernels import decimal128 as d128
        lo, hi = d128.limbs(s)
        i128 = np.empty((len(s), 2), dtype=np.int64)
        i128[:, 0] = lo.numpy()
        i128[:, 1] = hi.numpy()
        vbuf = pa.array(~mask).buffers()[1] if mask is not None else None
        return pa.Array.from_buffers(
            atype, len(s), [vbuf, pa.py_buffer(i128.tobytes())])
    if k == TypeKind.NULL:
        return pa.nulls(len(s))
    np_arr = s.data.numpy() if s.data.dtype not in (
        torch.uint16, torch.uint32, torch.uint64) else {
        torch.uint16: s.data.view(torch.int16).numpy().view(np.uint16),
        torch.uint32: s.data.view(torch.int32).numpy().view(np.uint32),
        torch.uint64: s.data.view(torch.int64).numpy().view(np.uint64),
    }[s.data.dtype]
    if k in (TypeKind.DATE, TypeKind.TIMESTAMP, TypeKind.DURATION,
             TypeKind.TIME):
        storage = pa.array(np_arr, mask=mask)
        return storage.cast(atype)
    return pa.array(np_arr, type=atype, mask=mask)
