"""Embedding distance kernels: batched via torch matmul/reduction on GPU
(MFMA-backed through rocBLAS for the batched GEMV shapes; ref:
/root/reference/src/daft-functions/src/distance/cosine.rs)."""
from __future__ import annotations

import torch

from ..schema import DataType
from ..series import Series


def _as_matrix(s: Series) -> torch.Tensor:
    n = len(s)
    sz = s.dtype.size
    return s.children[0].data.reshape(n, sz)


def cosine_distance_series(a: Series, b: Series) -> Series:
    n = max(len(a), len(b))
    if len(a) == 1:
        a = a.broadcast(n)
    if len(b) == 1:
        b = b.broadcast(n)
    ma = _as_matrix(a).to(torch.float32)
    mb = _as_matrix(b).to(torch.float32)
    dot = (ma * mb).sum(dim=1)
    na = ma.norm(dim=1)
    nb = mb.norm(dim=1)
    out = 1.0 - dot / (na * nb).clamp(min=1e-30)
    from ..kernels import _null_and
    return Series(a.name, DataType.float64(), data=out.to(torch.float64),
                  validity=_null_and(a.validity, b.validity))


def dot_series(a: Series, b: Series) -> Series:
    n = max(len(a), len(b))
    if len(a) == 1:
        a = a.broadcast(n)
    if len(b) == 1:
        b = b.broadcast(n)
    ma = _as_matrix(a).to(torch.float32)
    mb = _as_matrix(b).to(torch.float32)
    out = (ma * mb).sum(dim=1)
    from ..kernels import _null_and
    return Series(a.name, DataType.float64(), data=out.to(torch.float64),
                  validity=_null_and(a.validity, b.validity))


def l2_norm_series(a: Series) -> Series:
    ma = _as_matrix(a).to(torch.float32)
    out = ma.norm(dim=1)
    return Series(a.name, DataType.float64(), data=out.to(torch.float64),
                  validity=a.validity)
