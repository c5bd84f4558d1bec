"""Exact wide-decimal arithmetic (precision 19..38) on two int64 limb
tensors.

Physical storage for Decimal128(p>18, s) is struct<lo:int64, hi:int64>:
`lo` carries the low 64 bits of the scaled two's-complement i128 (its
int64 value is the BIT pattern — unsigned semantically), `hi` the high
64 bits (signed).  All arithmetic is plain torch int64 tensor ops with
explicit carries, so the SAME code is exact on CPU and on the GPU (no
custom kernel needed: wrapping int64 add/mul and bitwise ops are
device-portable).

Narrow decimals (p<=18) keep their scaled-int64 single-tensor layout
(schema.py to_physical); this module only ever sees wide ones.

ref semantics: /root/reference/src/daft-core/src/datatypes/ (Decimal128
logical type) and its i128 array ops; arrow Decimal128 little-endian
two-limb layout.
"""
from __future__ import annotations

from typing import List, Optional, Tuple

import torch

from ..schema import DataType, TypeKind

_MIN64 = -(1 << 63)
_MASK32 = 0xFFFFFFFF
_U64 = (1 << 64) - 1


def is_wide(dt: DataType) -> bool:
    return dt.kind == TypeKind.DECIMAL128 and dt.precision > 18


def limbs(s) -> Tuple[torch.Tensor, torch.Tensor]:
    """(lo, hi) int64 tensors of a wide-decimal Series."""
    return s.children[0].data, s.children[1].data


def make(name: str, dtype: DataType, lo: torch.Tensor, hi: torch.Tensor,
         validity: Optional[torch.Tensor] = None):
    from ..series import Series
    return Series(name, dtype,
                  children=[Series("lo", DataType.int64(), data=lo),
                            Series("hi", DataType.int64(), data=hi)],
                  validity=validity, length=lo.numel())


# ---------------------------------------------------------------------------
# python-int <-> limbs
# ---------------------------------------------------------------------------

def split_int(v: int) -> Tuple[int, int]:
    lo_u = v & _U64
    lo = lo_u - (1 << 64) if lo_u >= (1 << 63) else lo_u
    return lo, v >> 64          # python >> is arithmetic: correct hi


def join_int(lo: int, hi: int) -> int:
    return (hi << 64) | (lo & _U64)


def tensors_from_ints(ints: List[int], device="cpu"):
    los, his = [], []
    for v in ints:
        lo, hi = split_int(v)
        los.append(lo)
        his.append(hi)
    return (torch.tensor(los, dtype=torch.int64, device=device),
            torch.tensor(his, dtype=torch.int64, device=device))


def ints_from_tensors(lo: torch.Tensor, hi: torch.Tensor) -> List[int]:
    lo_l = lo.cpu().tolist()
    hi_l = hi.cpu().tolist()
    return [join_int(a, b) for a, b in zip(lo_l, hi_l)]


# ---------------------------------------------------------------------------
# 128-bit primitives (int64 tensors; lo is unsigned-semantics bits)
# ---------------------------------------------------------------------------

def _u_lt(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """Unsigned < on the int64 bit patterns (sign-flip trick)."""
    return (a ^ _MIN64) < (b ^ _MIN64)


def u_order_key(lo: torch.Tensor) -> torch.Tensor:
    """Order-preserving signed key for the unsigned lo limb."""
    return lo ^ _MIN64


def add128(alo, ahi, blo, bhi):
    lo = alo + blo                         # wrapping
    carry = _u_lt(lo, alo).to(torch.int64)
    return lo, ahi + bhi + carry


def sub128(alo, ahi, blo, bhi):
    lo = alo - blo
    borrow = _u_lt(alo, blo).to(torch.int64)
    return lo, ahi - bhi - borrow


def neg128(lo, hi):
    nlo = (~lo) + 1                        # wraps to 0 only when lo == 0
    carry = (lo == 0).to(torch.int64)
    return nlo, (~hi) + carry


def cmp128(alo, ahi, blo, bhi, op: str) -> torch.Tensor:
    if op == "eq":
        return (alo == blo) & (ahi == bhi)
    if op == "ne":
        return (alo != blo) | (ahi != bhi)
    lt = (ahi < bhi) | ((ahi == bhi) & _u_lt(alo, blo))
    if op == "lt":
        return lt
    if op == "ge":
        return ~lt
    gt = (ahi > bhi) | ((ahi == bhi) & _u_lt(blo, alo))
    if op == "gt":
        return gt
    if op == "le":
        return ~gt
    raise ValueError(op)


def mul128_small(lo, hi, c: int):
    """(lo, hi) * c for 0 <= c < 2^31 (wrapping at 128 bits)."""
    assert 0 <= c < (1 << 31)
    lo0 = lo & _MASK32                     # nonneg halves
    lo1 = (lo >> 32) & _MASK32
    p0 = lo0 * c
    p1 = lo1 * c + (p0 >> 32)              # nonneg, < 2^63
    out_lo = (p0 & _MASK32) | (p1 << 32)   # wrapping pack
    return out_lo, hi * c + (p1 >> 32)


def mul128_pow10(lo, hi, k: int):
    """(lo, hi) * 10^k via < 2^31 chunks."""
    while k >= 9:
        lo, hi = mul128_small(lo, hi, 10 ** 9)
        k -= 9
    if k:
        lo, hi = mul128_small(lo, hi, 10 ** k)
    return lo, hi


def _abs128(lo, hi):
    neg = hi < 0
    nlo, nhi = neg128(lo, hi)
    return (torch.where(neg, nlo, lo), torch.where(neg, nhi, hi), neg)


def _apply_sign(lo, hi, neg):
    nlo, nhi = neg128(lo, hi)
    return torch.where(neg, nlo, lo), torch.where(neg, nhi, hi)


def _udiv128_small(lo, hi, c: int):
    """Truncating unsigned division of a NONNEGATIVE (lo, hi) by
    0 < c < 2^31 (long division on 32-bit halves)."""
    q_hi = torch.div(hi, c, rounding_mode="floor")      # hi >= 0
    r = hi - q_hi * c                                   # 0 <= r < c
    lo1 = (lo >> 32) & _MASK32
    lo0 = lo & _MASK32
    a = r * (1 << 32) + lo1                             # < c * 2^32 <= 2^63
    q1 = torch.div(a, c, rounding_mode="floor")         # < 2^32
    r1 = a - q1 * c
    b = r1 * (1 << 32) + lo0
    q0 = torch.div(b, c, rounding_mode="floor")         # < 2^32
    q_lo = (q1 << 32) | q0
    return q_lo, q_hi


def divround128_pow10(lo, hi, k: int, round_half: bool = True):
    """(lo, hi) / 10^k: round-half-away-from-zero (default) or truncate
    toward zero.  Rounding adds 10^k/2 to |x| up front; the chunked
    truncating divisions compose exactly on non-negative values."""
    alo, ahi, neg = _abs128(lo, hi)
    if round_half:
        hlo, hhi = split_int(10 ** k // 2)
        alo, ahi = add128(alo, ahi, torch.full_like(alo, hlo),
                          torch.full_like(ahi, hhi))
    while k >= 9:
        alo, ahi = _udiv128_small(alo, ahi, 10 ** 9)
        k -= 9
    if k:
        alo, ahi = _udiv128_small(alo, ahi, 10 ** k)
    return _apply_sign(alo, ahi, neg)


def to_float64(lo, hi) -> torch.Tensor:
    """Approximate f64 value of the raw (unscaled) i128."""
    lo_u = (lo & ((1 << 62) - 1)).to(torch.float64) + \
        ((lo >> 62) & 3).to(torch.float64) * float(1 << 62)
    return hi.to(torch.float64) * float(1 << 64) + lo_u


def from_int64(v: torch.Tensor):
    """Sign-extend an int64 tensor into limbs."""
    return v, v >> 63


def mul128(alo, ahi, blo, bhi):
    """Full wrapping 128x128 -> low 128 bits (schoolbook on 32-bit
    halves; upper partial products beyond bit 127 are dropped, like
    native i128 multiplication)."""
    a0 = alo & _MASK32
    a1 = (alo >> 32) & _MASK32
    b0 = blo & _MASK32
    b1 = (blo >> 32) & _MASK32
    # low 64 bits with carry into the high limb.  32x32 partial products
    # can wrap int64's sign bit, so every right shift must be LOGICAL:
    # (t >> 32) & _MASK32
    p00 = a0 * b0
    p01 = a0 * b1
    p10 = a1 * b0
    mid = ((p00 >> 32) & _MASK32) + (p01 & _MASK32) + (p10 & _MASK32)
    lo = (p00 & _MASK32) | (mid << 32)
    carry = (mid >> 32) + ((p01 >> 32) & _MASK32) + ((p10 >> 32) & _MASK32)
    # high limb: a1*b1 + cross terms with the (signed) hi limbs, wrapping
    hi = a1 * b1 + carry + alo * bhi + ahi * blo
    return lo, hi


# ---------------------------------------------------------------------------
# reductions (exact SUM via 32-bit half accumulation; lexicographic
# MIN/MAX) — `seg` is a callable (vals_int64) -> per-group int64 sums,
# so the same code serves global and grouped aggregation.
# ---------------------------------------------------------------------------

def sum128(lo, hi, seg):
    """seg(v) must return exact int64 sums (n < 2^31 rows per call)."""
    s0 = seg(lo & _MASK32)
    s1 = seg((lo >> 32) & _MASK32)
    sh = seg(hi)                           # wrap on true i128 overflow
    carry = (s0 >> 32) + s1                # nonneg
    out_lo = (s0 & _MASK32) | ((carry & _MASK32) << 32)
    return out_lo, sh + (carry >> 32)
