"""MinHash signatures over word n-grams (ref: /root/reference/src/
daft-minhash/src/lib.rs:279-330 — SIMD permutations (a*h+b) mod 2^61-1 with
running min; here one 64-lane wave per row on GPU)."""
from __future__ import annotations

import numpy as np
import torch

from ..schema import DataType
from ..series import Series
from ..kernels import _is_gpu, native_required

MERSENNE61 = (1 << 61) - 1


def _perms(num_hashes: int, seed: int):
    rng = np.random.RandomState(seed)
    a = rng.randint(1, MERSENNE61, size=num_hashes, dtype=np.uint64)
    b = rng.randint(0, MERSENNE61, size=num_hashes, dtype=np.uint64)
    return a, b


def minhash_series(s: Series, num_hashes: int, ngram_size: int,
                   seed: int) -> Series:
    a, b = _perms(num_hashes, seed)
    out_dt = DataType.fixed_size_list(DataType.uint32(), num_hashes)
    if _is_gpu(s):
        pa = torch.from_numpy(a.view(np.int64)).to(s.device)
        pb = torch.from_numpy(b.view(np.int64)).to(s.device)
        flat = native_required().minhash(s.offsets, s.data, num_hashes,
                                         ngram_size, pa, pb)
        child = Series("item", DataType.uint32(),
                       data=flat.view(torch.uint32))
        return Series(s.name, out_dt, children=[child],
                      validity=s.validity, length=len(s))
    # CPU fallback: same algorithm
    from ..kernels.rowops import _fnv1a_bytes
    from .. import kernels
    vals = s.to_pylist()
    out = np.full((len(vals), num_hashes), 0xFFFFFFFF, dtype=np.uint32)
    for i, v in enumerate(vals):
        if v is None:
            continue
        words = v.split(" ") if isinstance(v, str) else v.decode().split(" ")
        words = [w for w in words if w]
        grams = [" ".join(words[j:j + ngram_size])
                 for j in range(max(0, len(words) - ngram_size + 1))]
        if not grams:
            continue
        mins = np.full(num_hashes, np.iinfo(np.uint64).max, dtype=np.uint64)
        for g in grams:
            h = np.uint64(_hash_bytes_py(g.encode()) & MERSENNE61)
            with np.errstate(over="ignore"):
                v64 = (_mulmod61_np(a, h) + b)
            v64 = np.where(v64 >= MERSENNE61, v64 - MERSENNE61, v64)
            mins = np.minimum(mins, v64)
        out[i] = (mins & np.uint64(0xFFFFFFFF)).astype(np.uint32)
    child = Series("item", DataType.uint32(),
                   data=torch.from_numpy(out.reshape(-1).view(np.int32))
                   .view(torch.uint32))
    return Series(s.name, out_dt, children=[child], validity=s.validity,
                  length=len(s))


def _hash_bytes_py(p: bytes) -> int:
    """Match csrc/common.h hash_bytes_dev."""
    def splitmix64(x):
        x = (x + 0x9E3779B97F4A7C15) & 0xFFFFFFFFFFFFFFFF
        x = ((x ^ (x >> 30)) * 0xBF58476D1CE4E5B9) & 0xFFFFFFFFFFFFFFFF
        x = ((x ^ (x >> 27)) * 0x94D049BB133111EB) & 0xFFFFFFFFFFFFFFFF
        return x ^ (x >> 31)
    h = (0xcbf29ce484222325 ^ len(p)) & 0xFFFFFFFFFFFFFFFF
    i = 0
    while i + 8 <= len(p):
        w = int.from_bytes(p[i:i + 8], "little")
        h = splitmix64(h ^ w)
        i += 8
    if i < len(p):
        last = int.from_bytes(p[i:], "little")
        h = splitmix64(h ^ last)
    return h


def _mulmod61_np(a: np.ndarray, h: np.uint64) -> np.ndarray:
    out = np.empty_like(a)
    hv = int(h)
    for i, av in enumerate(a):
        out[i] = (int(av) * hv) % MERSENNE61
    return out


def simhash_series(s: Series, ngram_size: int) -> Series:
    """64-bit SimHash fingerprint over byte n-grams (capability of
    /root/reference/src/daft-functions/src/simhash.rs:15-42: per-bit
    majority vote over ngram hashes; our hash is csrc/common.h
    hash_bytes_dev).  Rows shorter than ngram_size fingerprint to 0."""
    n = len(s)
    if _is_gpu(s):
        out = native_required().simhash(s.offsets, s.data, ngram_size)
        return Series(s.name, DataType.uint64(),
                      data=out.view(torch.uint64), validity=s.validity)
    vals = s.to_pylist()
    out = np.zeros(n, dtype=np.uint64)
    for i, v in enumerate(vals):
        if v is None:
            continue
        b = v.encode() if isinstance(v, str) else bytes(v)
        m = len(b) - ngram_size + 1
        if m <= 0:
            continue
        w = np.zeros(64, dtype=np.int64)
        for j in range(m):
            h = _hash_bytes_py(b[j:j + ngram_size])
            bits = (np.uint64(h) >> np.arange(64, dtype=np.uint64)) \
                & np.uint64(1)
            w += np.where(bits.astype(bool), 1, -1)
        fp = 0
        for i2 in range(64):
            if w[i2] > 0:
                fp |= 1 << i2
        out[i] = np.uint64(fp)
    return Series(s.name, DataType.uint64(),
                  data=torch.from_numpy(out.view(np.int64))
                  .view(torch.uint64), validity=s.validity)
