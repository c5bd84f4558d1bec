"""Row-wise kernels: hashing, hash groupby, hash join, multi-key argsort,
hash partitioning.

GPU path: hand-written HIP/CDNA4 kernels in csrc/ (bucket-chained hash
tables in HBM, LDS-histogram radix sort, one-thread-per-row xxhash-style
row hashing) — the MI355X-native equivalents of the reference's
daft-core/src/kernels/hashing.rs, daft-recordbatch/src/probeable/ and
daft-core/src/array/ops/sort.rs.

CPU path: numpy/dict fallbacks, used only by the no-GPU test tier.
"""
from __future__ import annotations

from typing import List, Optional, Sequence, Tuple

import numpy as np
import torch

from ..schema import DataType, TypeKind
from ..series import Series
from . import _descs, _is_gpu, native_required, load_native

_SIGN64 = -0x8000000000000000  # 1 << 63 as int64


# ---------------------------------------------------------------------------
# row hashing
# ---------------------------------------------------------------------------

def hash_columns(series: Sequence[Series], seed: int = 0) -> torch.Tensor:
    """64-bit combined row hash over the given key columns (int64 bit-pattern)."""
    for s in series:
        if s.dtype.is_decimal() and s.children:
            raise NotImplementedError(
                "group/join/partition keys of wide decimals (p>18) are "
                "not supported — cast to a p<=18 decimal, integer or "
                "string key first")
    if _is_gpu(series[0]):
        tags, datas, offs, vals = _descs(series)
        return native_required().hash_rows(tags, datas, offs, vals,
                                           len(series[0]), seed)
    return _cpu_hash_columns(series, seed)


def _splitmix64_np(x: np.ndarray) -> np.ndarray:
    x = x.astype(np.uint64, copy=True)
    with np.errstate(over="ignore"):
        x += np.uint64(0x9E3779B97F4A7C15)
        z = x
        z = (z ^ (z >> np.uint64(30))) * np.uint64(0xBF58476D1CE4E5B9)
        z = (z ^ (z >> np.uint64(27))) * np.uint64(0x94D049BB133111EB)
        z = z ^ (z >> np.uint64(31))
    return z


def _fnv1a_bytes(b: bytes) -> int:
    h = 0xcbf29ce484222325
    for c in b:
        h ^= c
        h = (h * 0x100000001b3) & 0xFFFFFFFFFFFFFFFF
    return h


_NULL_HASH = np.uint64(0x9E3779B97F4A7C15)



def _phys_float(dt: DataType) -> bool:
    """True when the physical storage is floating point (decimals with
    p <= 18 store scaled int64 and take the exact integer paths)."""
    return dt.to_physical().kind in (TypeKind.FLOAT32, TypeKind.FLOAT64)

def _cpu_hash_columns(series: Sequence[Series], seed: int) -> torch.Tensor:
    n = len(series[0])
    acc = np.full(n, np.uint64(seed) + np.uint64(0x8445D61A4E774912),
                  dtype=np.uint64)
    for s in series:
        if s.is_dict():
            s = s.dict_decode()
        k = s.dtype.kind
        if k in (TypeKind.STRING, TypeKind.BINARY):
            off = s.offsets.numpy()
            buf = s.data.numpy().tobytes()
            h = np.fromiter(
                (_fnv1a_bytes(buf[off[i]:off[i + 1]]) for i in range(n)),
                dtype=np.uint64, count=n)
        else:
            d = s.data
            if d.dtype == torch.bool:
                d = d.to(torch.int8)
            if d.dtype in (torch.float32, torch.float64):
                dd = d.to(torch.float64)
                # normalize -0.0 and NaN for hash equality
                dd = torch.where(dd == 0.0, torch.zeros_like(dd), dd)
                dd = torch.where(torch.isnan(dd), torch.full_like(dd, float("nan")), dd)
                raw = dd.view(torch.int64).numpy().astype(np.int64)
            else:
                raw = d.to(torch.int64).numpy() if d.dtype != torch.uint64 \
                    else d.view(torch.int64).numpy()
            h = _splitmix64_np(raw.view(np.uint64) if raw.dtype == np.int64
                               else raw.astype(np.uint64))
        if s.validity is not None:
            v = s.validity.numpy()
            h = np.where(v, h, _NULL_HASH)
        with np.errstate(over="ignore"):
            acc = (acc * np.uint64(0x9E3779B97F4A7C15)) ^ _splitmix64_np(h)
    return torch.from_numpy(acc.view(np.int64))


# ---------------------------------------------------------------------------
# key tuples for CPU dict fallbacks
# ---------------------------------------------------------------------------

def _cpu_group_codes(s: Series):
    """Vectorized per-column dense codes for the CPU groupby fallback:
    (codes int64 in [0, card), card), or None for nested types."""
    n = len(s)
    valid = s.validity.numpy() if s.validity is not None else None
    k = s.dtype.kind
    dense = None  # (codes, card) already bounded: skip the unique pass
    if s.is_dict():
        dense = (s.data.to(torch.int64).numpy(),
                 max(len(s.children[0]), 1))
    elif k in (TypeKind.STRING, TypeKind.BINARY):
        from .. import arrow_interop
        d = arrow_interop.to_arrow_array(s)
        if d.null_count:
            d = d.fill_null(b"" if k == TypeKind.BINARY else "")
        enc = d.dictionary_encode()
        dense = (np.asarray(enc.indices).astype(np.int64),
                 max(len(enc.dictionary), 1))
    if dense is not None:
        codes, card = dense
        if valid is not None:
            codes = np.where(valid, codes, card)
            card += 1
        return codes, card
    if _phys_float(s.dtype):
        d = s.data.to(torch.float64)
        d = torch.where(d == 0.0, torch.zeros_like(d), d)
        d = torch.where(torch.isnan(d), torch.full_like(d, float("nan")), d)
        vals = d.view(torch.int64).numpy()
    elif s.data is not None and s.dtype.is_fixed_width():
        d = s.data
        if d.dtype == torch.bool:
            d = d.to(torch.int8)
        vals = (d.view(torch.int64) if d.dtype == torch.uint64
                else d.to(torch.int64)).numpy()
        if n and d.dtype != torch.uint64:
            mn, mx = int(vals.min()), int(vals.max())
            rng = mx - mn + 1
            if 0 < rng <= max(2 * n, 1 << 16):
                codes = vals - mn
                if valid is not None:
                    codes = np.where(valid, codes, rng)
                    rng += 1
                return codes, rng
    else:
        return None
    if valid is not None and n:
        vals = np.where(valid, vals, vals.flat[0])
    uniq, inv = np.unique(vals, return_inverse=True)
    codes = inv.astype(np.int64)
    card = len(uniq)
    if valid is not None:
        codes = np.where(valid, codes, card)
        card += 1
    return codes, max(card, 1)


def _cpu_groupby_vectorized(keys: Sequence[Series]):
    """numpy groupby: per-column dense codes packed into one int64, then a
    single np.unique.  Returns (gids, reps) or None when a column type is
    unsupported or the packed key range overflows."""
    packed = None
    total = 1
    for s in keys:
        enc = _cpu_group_codes(s)
        if enc is None:
            return None
        codes, card = enc
        if total * card >= (1 << 62):
            return None
        packed = codes if packed is None else packed * card + codes
        total *= card
    n = len(packed)
    # hash-based dedup (pyarrow) beats np.unique's O(n log n) sort; group
    # ids come out in first-appearance order, matching the dict fallback
    import pyarrow as pa
    enc = pa.array(packed, type=pa.int64()).dictionary_encode()
    gids = torch.from_numpy(np.asarray(enc.indices).astype(np.int64, copy=False))
    card = len(enc.dictionary)
    reps = torch.full((card,), n, dtype=torch.int64)
    reps.scatter_reduce_(0, gids, torch.arange(n, dtype=torch.int64),
                         reduce="amin", include_self=True)
    return gids, reps


def _cpu_key_rows(series: Sequence[Series]) -> list:
    cols = []
    for s in series:
        vals = s.to_pylist()
        if _phys_float(s.dtype):
            vals = [None if v is None
                    else (0.0 if v == 0.0 else ("nan" if v != v else v))
                    for v in vals]
        cols.append(vals)
    return list(zip(*cols)) if cols else []


# ---------------------------------------------------------------------------
# groupby: rows -> dense group ids (+ representative row per group)
# ---------------------------------------------------------------------------

def groupby(keys: Sequence[Series]) -> Tuple[torch.Tensor, torch.Tensor]:
    """Return (group_ids[n] int64, rep_idx[num_groups] int64).

    Group ids are dense [0, num_groups); rep_idx[g] is a row index whose key
    values represent group g (for gathering output key columns).
    """
    if _is_gpu(keys[0]):
        dense = _dense_range_groupby(keys)
        if dense is not None:
            return dense
        hashes = hash_columns(keys)
        tags, datas, offs, vals = _descs(keys)
        return native_required().groupby(hashes, tags, datas, offs, vals)
    vec = _cpu_groupby_vectorized(keys)
    if vec is not None:
        return vec
    rows = _cpu_key_rows(keys)
    seen = {}
    gids = np.empty(len(rows), dtype=np.int64)
    reps: List[int] = []
    for i, r in enumerate(rows):
        g = seen.get(r)
        if g is None:
            g = len(reps)
            seen[r] = g
            reps.append(i)
        gids[i] = g
    return (torch.from_numpy(gids),
            torch.tensor(reps, dtype=torch.int64))


_DENSE_RANGE_LIMIT = 1 << 28  # map memory cap: 256M entries (2 GB int64)


def _dense_range_groupby(keys):
    """Dense-range fast path: when every key is a no-null integer (or dict
    code / date / bool) and the product of value ranges is small, pack keys
    into one index and densify with two scatter passes — no hash table.
    The common analytics shape (dense surrogate keys, dict codes)."""
    packed = None
    rng_prod = 1
    n = len(keys[0])
    if n == 0:
        return None
    for s in keys:
        if s.validity is not None:
            return None
        if s.is_dict():
            k = s.data.to(torch.int64)
        elif s.data is not None and s.dtype.is_fixed_width() and \
                s.data.dtype != torch.uint64 and (
                s.dtype.is_integer() or s.dtype.is_boolean() or
                s.dtype.kind in (TypeKind.DATE,)):
            k = s.data.to(torch.int64)
        else:
            return None
        mn = int(k.min().item())
        mx = int(k.max().item())
        rng = mx - mn + 1
        if rng <= 0 or rng_prod * rng > _DENSE_RANGE_LIMIT:
            return None
        packed = (k - mn) if packed is None else packed * rng + (k - mn)
        rng_prod *= rng
    first = native_required().dense_first_index(packed, rng_prod, n)
    present = first < n
    ids_map = torch.cumsum(present.to(torch.int64), 0) - 1
    gids = ids_map[packed]
    reps = first[present]
    return gids, reps


_SEG_MIN_GROUPS = 1 << 16   # atomics win below this; clustered-key wins above


def _segmented_grouped_agg(gid, num_groups, d, validity, op, dev):
    """Sorted dense gids (orderkey-clustered aggregations, q21/q18):
    torch.segment_reduce over searchsorted boundaries instead of scattered
    global atomics.  Returns (out, cnt) or None when not applicable."""
    n = gid.numel()
    if num_groups < _SEG_MIN_GROUPS or n < 2:
        return None
    if not bool((gid[1:] >= gid[:-1]).all().item()):
        return None
    int_in = not d.dtype.is_floating_point
    if int_in:
        # segment_reduce has no CUDA Long kernel: run through f64 when
        # values stay exactly representable, else fall back to the
        # hash-table kernel
        if d.numel() and int(d.abs().max().item()) >= (1 << 53):
            return None
        d = d.to(torch.float64)
    starts = torch.searchsorted(gid, torch.arange(
        num_groups, dtype=gid.dtype, device=dev))
    bounds = torch.cat([starts, torch.tensor([n], dtype=starts.dtype,
                                             device=dev)])
    lengths = torch.diff(bounds)
    if validity is None:
        cnt = lengths.to(torch.int64)
        dd = d
    else:
        cnt = torch.segment_reduce(validity.to(torch.float32), "sum",
                                   lengths=lengths).to(torch.int64)
        if op == "sum":
            dd = torch.where(validity, d, torch.zeros_like(d))
        else:
            fill = float("inf") if op == "min" else float("-inf")
            dd = torch.where(validity, d, torch.full_like(d, fill))
    out = torch.segment_reduce(dd, op, lengths=lengths,
                               initial=0.0 if op == "sum" else
                               (float("inf") if op == "min"
                                else float("-inf")))
    if int_in:
        out = out.to(torch.int64)
    return out, cnt


def grouped_agg(group_ids: torch.Tensor, num_groups: int, values: Series,
                op: str) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """Aggregate values per group.  op in {sum,min,max,count,count_valid,sum_sq}.

    Returns (agg_data[num_groups], valid_count[num_groups] or None).
    `count` counts all rows; `count_valid` counts non-null rows.
    """
    dev = values.device
    gid = group_ids
    if op in ("count", "count_valid"):
        validity = values.validity if op == "count_valid" else None
        if _is_gpu(values):
            vmask = validity if validity is not None else \
                torch.empty(0, dtype=torch.bool, device=dev)
            out = native_required().grouped_count(gid, num_groups, vmask)
            return out, None
        ones = torch.ones(len(values), dtype=torch.int64, device=dev)
        if validity is not None:
            ones = ones * validity.to(torch.int64)
        out = torch.zeros(num_groups, dtype=torch.int64, device=dev)
        out.scatter_add_(0, gid, ones)
        return out, None
    data = values.data
    validity = values.validity

    if _phys_float(values.dtype):
        wdt = torch.float64
    elif values.dtype.kind == TypeKind.FLOAT32:
        wdt = torch.float64
    else:
        wdt = torch.int64
    d = data.to(wdt)
    if op == "sum_sq":
        d = d.to(torch.float64)
        d = d * d
        wdt = torch.float64
        op = "sum"

    if _is_gpu(values) and op in ("sum", "min", "max"):
        seg = _segmented_grouped_agg(gid, num_groups, d, validity, op, dev)
        if seg is not None:
            return seg
        vmask = validity if validity is not None else \
            torch.empty(0, dtype=torch.bool, device=dev)
        out, cnt = native_required().grouped_agg(gid, num_groups, d, vmask, op)
        if validity is None and cnt is not None:
            # kernel counted every row (no validity): cnt is still exact
            pass
        return out, cnt

    # CPU fallback via scatter_reduce
    vmask = validity if validity is not None else \
        torch.ones(len(values), dtype=torch.bool, device=dev)
    cnt = torch.zeros(num_groups, dtype=torch.int64, device=dev)
    cnt.scatter_add_(0, gid, vmask.to(torch.int64))
    if op == "sum":
        out = torch.zeros(num_groups, dtype=wdt, device=dev)
        dd = torch.where(vmask, d, torch.zeros_like(d))
        out.scatter_add_(0, gid, dd)
        return out, cnt
    if op in ("min", "max"):
        if wdt == torch.float64:
            fill = float("inf") if op == "min" else float("-inf")
        else:
            fill = (2**63 - 1) if op == "min" else -(2**63)
        dd = torch.where(vmask, d, torch.full_like(d, fill))
        out = torch.full((num_groups,), fill, dtype=wdt, device=dev)
        out.scatter_reduce_(0, gid, dd, reduce="amin" if op == "min" else "amax")
        return out, cnt
    raise ValueError(f"unknown agg op {op}")


# ---------------------------------------------------------------------------
# hash join
# ---------------------------------------------------------------------------

def join(left_keys: Sequence[Series], right_keys: Sequence[Series],
         how: str) -> Tuple[torch.Tensor, torch.Tensor]:
    """Hash join; returns (left_idx, right_idx) int64 gather maps.

    -1 entries mean "null row" (unmatched side of outer joins).
    how in {inner,left,right,outer,semi,anti}.  For semi/anti only left_idx
    is meaningful (right_idx is empty).
    Build side is the RIGHT side (callers put the smaller input right).
    """
    from ..schema import supertype
    lk2, rk2 = [], []
    for l, r in zip(left_keys, right_keys):
        if l.is_dict() or r.is_dict():
            l, r = _align_dict_keys(l, r)
        st = supertype(l.dtype, r.dtype)
        lk2.append(l.cast(st) if l.dtype != st else l)
        rk2.append(r.cast(st) if r.dtype != st else r)
    left_keys, right_keys = lk2, rk2
    packed = _pack_int_keys(left_keys, right_keys)
    if packed is not None:
        left_keys, right_keys = packed
    dense = _dense_key_join(left_keys, right_keys, how)
    if dense is not None:
        return dense
    if _is_gpu(left_keys[0]):
        return _gpu_join(left_keys, right_keys, how)
    return _cpu_join(left_keys, right_keys, how)


def _pack_int_keys(lk, rk):
    """Fold a multi-column integer equi-key into ONE synthetic int64
    column per side ((a-mnA)*rngB + (b-mnB)...): the probe then hashes
    and compares a single word — and composite PK joins (e.g. partsupp's
    (partkey, suppkey)) become eligible for the direct-address path.
    Null semantics preserved: the packed column carries the AND of the
    per-column validities (null keys never match)."""
    if len(lk) < 2:
        return None
    mins, rngs = [], []
    for i in range(len(lk)):
        for s in (lk[i], rk[i]):
            if s.is_dict() or s.data is None or s.data.dtype == \
                    torch.uint64 or not (
                    s.dtype.is_integer() or
                    (s.dtype.is_decimal() and
                     s.data.dtype == torch.int64) or
                    s.dtype.kind == TypeKind.DATE):
                return None
            if len(s) == 0:
                return None
        lo = min(int(lk[i].data.min().item()),
                 int(rk[i].data.min().item()))
        hi = max(int(lk[i].data.max().item()),
                 int(rk[i].data.max().item()))
        mins.append(lo)
        rngs.append(hi - lo + 1)
    total = 1
    for r in rngs:
        total *= r
        if total >= (1 << 62):
            return None

    def pack(cols):
        acc = None
        validity = None
        for s, mn, rng in zip(cols, mins, rngs):
            d = s.data.to(torch.int64) - mn
            acc = d if acc is None else acc * rng + d
            if s.validity is not None:
                validity = s.validity if validity is None \
                    else (validity & s.validity)
        return [Series(cols[0].name, DataType.int64(), data=acc,
                       validity=validity)]
    return pack(lk), pack(rk)


_DENSE_JOIN_LIMIT = 1 << 31  # direct table cap: 2G slots (8 GB int32)


def _dense_key_join(lk, rk, how):
    """Direct-address join: when the build side is a single no-null
    integer key whose values are unique and densely ranged (the PK-join
    shape — orderkey/partkey/suppkey in TPC-H), replace the hash table
    with an addressed row array.  Probe is then one gather + a compare:
    no hashing, no chain walks, no count/fill passes.  Returns None to
    fall back to the bucket-chain hash join."""
    if len(lk) != 1 or len(rk) != 1:
        return None
    l, r = lk[0], rk[0]
    for s in (l, r):
        if s.is_dict() or s.data is None or not (
                s.dtype.is_integer() or
                (s.dtype.is_decimal() and s.data.dtype == torch.int64) or
                s.dtype.kind in (TypeKind.DATE,)):
            return None
        if s.data.dtype == torch.uint64:
            return None
    if r.validity is not None and bool((~r.validity).any()):
        return None
    n_r = len(r)
    n_l = len(l)
    if n_r == 0 or n_l == 0:
        return None
    dev = r.device
    rdata = r.data.to(torch.int64)
    mn = int(rdata.min().item())
    mx = int(rdata.max().item())
    rng = mx - mn + 1
    # 16x sparsity still wins: the direct table costs rng x 4B of HBM
    # (transient) and one memset, vs a chained hash build + 2-pass probe
    if rng <= 0 or rng > max(16 * n_r, 1 << 20) or \
            rng > _DENSE_JOIN_LIMIT:
        return None
    idx_t = torch.int32 if n_r < (1 << 31) else torch.int64
    table = torch.full((rng,), -1, dtype=idx_t, device=dev)
    rows = torch.arange(n_r, dtype=idx_t, device=dev)
    table[rdata - mn] = rows
    if how not in ("semi", "anti") and \
            not bool((table[rdata - mn] == rows).all()):
        return None  # duplicate build keys: hash join handles fan-out
    # (semi/anti only test existence, so duplicate build keys are fine)
    ldata = l.data.to(torch.int64)
    in_rng = (ldata >= mn) & (ldata <= mx)
    if l.validity is not None:
        in_rng &= l.validity
    row = table[(ldata - mn).clamp(min=0, max=rng - 1)].to(torch.int64)
    hit = in_rng & (row >= 0)
    ar = torch.arange(n_l, dtype=torch.int64, device=dev)
    if how == "semi":
        return ar[hit], torch.zeros(0, dtype=torch.int64, device=dev)
    if how == "anti":
        return ar[~hit], torch.zeros(0, dtype=torch.int64, device=dev)
    if how == "inner":
        lidx = ar[hit]
        return lidx, row[lidx]
    if how == "left":
        return ar, torch.where(hit, row, torch.full_like(row, -1))
    # right / outer: matched build rows + appended unmatched
    lidx = ar[hit] if how == "right" else ar
    ridx = row[lidx] if how == "right" else \
        torch.where(hit, row, torch.full_like(row, -1))
    matched = torch.zeros(n_r, dtype=torch.bool, device=dev)
    matched[row[hit]] = True
    unmatched = torch.nonzero(~matched).reshape(-1)
    if unmatched.numel():
        lidx = torch.cat([lidx, torch.full_like(unmatched, -1)])
        ridx = torch.cat([ridx, unmatched])
    return lidx, ridx


def _align_dict_keys(l: Series, r: Series):
    """Re-encode the right dict column into the left vocab so code equality
    matches string equality (vocab entries are distinct by construction)."""
    from . import _same_vocab
    if l.is_dict() and r.is_dict():
        lv, rv = l.children[0], r.children[0]
        if _same_vocab(lv, rv):
            return l, r
        lmap = {v: i for i, v in enumerate(lv.to_pylist())}
        rvals = rv.to_pylist()
        remap = torch.tensor([lmap.get(v, -1) for v in rvals],
                             dtype=torch.int32, device=r.device)
        new_codes = remap[r.data.to(torch.int64)]
        validity = (new_codes >= 0)
        if r.validity is not None:
            validity = validity & r.validity
        return l, Series.make_dict(r.name, lv, new_codes.clamp(min=0),
                                   validity)
    return l.dict_decode(), r.dict_decode()


def _gpu_join(lk, rk, how):
    nat = native_required()
    rh = hash_columns(rk)
    lh = hash_columns(lk)
    table, nxt = nat.join_build(rh)
    tags_l, data_l, off_l, val_l = _descs(lk)
    tags_r, data_r, off_r, val_r = _descs(rk)
    mode = {"inner": 0, "left": 1, "semi": 2, "anti": 3,
            "right": 0, "outer": 1}[how]
    lidx, ridx, matched = nat.join_probe(
        table, nxt, lh, rh, tags_l, data_l, off_l, val_l,
        tags_r, data_r, off_r, val_r, mode)
    if how in ("right", "outer"):
        unmatched = torch.nonzero(~matched.to(torch.bool)).reshape(-1)
        if unmatched.numel():
            lidx = torch.cat([lidx, torch.full_like(unmatched, -1)])
            ridx = torch.cat([ridx, unmatched])
    return lidx, ridx


def _cpu_join_vectorized(lk, rk, how):
    """numpy sort-merge join on packed per-row codes (shared encodings via
    a concatenated unique pass).  None when a key type is unsupported."""
    from . import concat as _concat
    nl, nr = len(lk[0]), len(rk[0])
    packed = None
    total = 1
    lvalid = np.ones(nl, dtype=bool)
    rvalid = np.ones(nr, dtype=bool)
    for l, r in zip(lk, rk):
        enc = _cpu_group_codes(_concat([l, r]))
        if enc is None:
            return None
        codes, card = enc
        if total * card >= (1 << 62):
            return None
        packed = codes if packed is None else packed * card + codes
        total *= card
        if l.validity is not None:
            lvalid &= l.validity.numpy()
        if r.validity is not None:
            rvalid &= r.validity.numpy()
    lcodes, rcodes = packed[:nl].copy(), packed[nl:].copy()
    # null keys never match (SQL semantics): sentinel below any real code
    lcodes[~lvalid] = -2
    rcodes[~rvalid] = -1
    rs = np.argsort(rcodes, kind="stable")
    rsorted = rcodes[rs]
    lo = np.searchsorted(rsorted, lcodes, "left")
    hi = np.searchsorted(rsorted, lcodes, "right")
    counts = hi - lo
    if how == "semi":
        return (torch.from_numpy(np.flatnonzero(counts > 0)),
                torch.zeros(0, dtype=torch.int64))
    if how == "anti":
        return (torch.from_numpy(np.flatnonzero(counts == 0)),
                torch.zeros(0, dtype=torch.int64))
    tot = int(counts.sum())
    li = np.repeat(np.arange(nl, dtype=np.int64), counts)
    start = np.cumsum(counts) - counts
    pos = np.arange(tot, dtype=np.int64) - np.repeat(start, counts) \
        + np.repeat(lo, counts)
    ri = rs[pos] if tot else np.zeros(0, dtype=np.int64)
    if how in ("left", "outer"):
        miss = np.flatnonzero(counts == 0)
        li = np.concatenate([li, miss])
        ri = np.concatenate([ri, np.full(len(miss), -1, dtype=np.int64)])
    if how in ("right", "outer"):
        matched = np.zeros(nr, dtype=bool)
        matched[ri[ri >= 0]] = True
        miss = np.flatnonzero(~matched)
        li = np.concatenate([li, np.full(len(miss), -1, dtype=np.int64)])
        ri = np.concatenate([ri, miss])
    return (torch.from_numpy(li.astype(np.int64, copy=False)),
            torch.from_numpy(ri.astype(np.int64, copy=False)))


def _cpu_join(lk, rk, how):
    vec = _cpu_join_vectorized(lk, rk, how)
    if vec is not None:
        return vec
    lrows = _cpu_key_rows(lk)
    rrows = _cpu_key_rows(rk)
    table = {}
    for j, r in enumerate(rrows):
        if any(v is None for v in r):
            continue  # null keys never match
        table.setdefault(r, []).append(j)
    li: List[int] = []
    ri: List[int] = []
    matched_r = np.zeros(len(rrows), dtype=bool)
    for i, r in enumerate(lrows):
        ms = table.get(r) if not any(v is None for v in r) else None
        if ms:
            if how == "semi":
                li.append(i)
                continue
            if how == "anti":
                continue
            for j in ms:
                li.append(i)
                ri.append(j)
                matched_r[j] = True
        else:
            if how == "anti":
                li.append(i)
            elif how in ("left", "outer"):
                li.append(i)
                ri.append(-1)
    if how in ("right", "outer"):
        for j in np.nonzero(~matched_r)[0]:
            li.append(-1)
            ri.append(int(j))
    lidx = torch.tensor(li, dtype=torch.int64)
    ridx = torch.tensor(ri, dtype=torch.int64) if how not in ("semi", "anti") \
        else torch.zeros(0, dtype=torch.int64)
    return lidx, ridx


# ---------------------------------------------------------------------------
# multi-key argsort (stable LSD radix on order-preserving u64 keys)
# ---------------------------------------------------------------------------

def _order_key_u64(s: Series) -> torch.Tensor:
    """Order-preserving u64 (as int64 bit pattern) for fixed-width dtypes."""
    d = s.data
    k = s.dtype.kind
    if k == TypeKind.BOOL:
        return d.to(torch.int64)
    if _phys_float(s.dtype):
        bits = d.to(torch.float64).view(torch.int64)
        neg = bits < 0
        flipped = torch.where(neg, ~bits, bits ^ _SIGN64)
        return flipped
    if s.dtype.is_unsigned_integer():
        if d.dtype == torch.uint64:
            return d.view(torch.int64)
        return d.to(torch.int64)
    # signed ints / date / timestamp / time / duration
    return d.to(torch.int64) ^ _SIGN64


def _string_chunk_key(s: Series, chunk: int,
                      idx: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Big-endian u64 of bytes [8c, 8c+8), zero padded — lexicographic LSD chunks."""
    if _is_gpu(s):
        return native_required().string_chunk_key(s.offsets, s.data, chunk)
    off = s.offsets.numpy()
    buf = s.data.numpy().tobytes()
    n = len(s)
    out = np.zeros(n, dtype=np.uint64)
    base = 8 * chunk
    for i in range(n):
        a, b = off[i], off[i + 1]
        if a + base >= b:
            continue
        seg = buf[a + base:min(a + base + 8, b)]
        out[i] = int.from_bytes(seg.ljust(8, b"\0"), "big")
    return torch.from_numpy(out.view(np.int64))


def _max_strlen(s: Series) -> int:
    if len(s) == 0:
        return 0
    lens = s.offsets[1:] - s.offsets[:-1]
    return int(lens.max().item())


def _stable_sort_perm_by(keys: torch.Tensor) -> torch.Tensor:
    """Stable ascending argsort of u64-encoded keys (int64 bit pattern,
    compared as unsigned)."""
    if keys.is_cuda:
        return native_required().radix_argsort(keys)
    u = keys.numpy().view(np.uint64)
    return torch.from_numpy(np.argsort(u, kind="stable").astype(np.int64))


def _key_bits(s: Series):
    """(biased int64 key ascending-ordered, bit width) or None when the
    column can't compose into a packed radix key."""
    if s.is_dict():
        vocab = s.children[0]
        vperm = argsort_multi([vocab], [False], [False])
        rank = torch.empty(len(vocab), dtype=torch.int64, device=s.device)
        rank[vperm] = torch.arange(len(vocab), dtype=torch.int64,
                                   device=s.device)
        k = rank[s.data.to(torch.int64)]
        card = max(len(vocab), 1)
        return k, max(card - 1, 1).bit_length()
    dt = s.dtype
    if not (dt.is_integer() or dt.is_boolean() or dt.is_temporal()):
        return None
    if s.data is None or s.data.dtype == torch.uint64:
        return None
    v = s.data.to(torch.int64)
    if s.validity is not None:
        v = torch.where(s.validity, v, torch.zeros_like(v))
    lo = int(v.min().item()) if v.numel() else 0
    hi = int(v.max().item()) if v.numel() else 0
    span = hi - lo
    if span < 0 or span >= (1 << 62):
        return None
    return v - lo, max(span, 1).bit_length()


def _try_composed_argsort(keys, descending, nulls_first):
    """Pack every key (plus a null bit where needed) into ONE u64 and run
    a single radix sort instead of one full sort per key (SURVEY §2.5
    key-composition; the per-key LSD loop below is the fallback)."""
    if len(keys) < 2:
        return None
    parts = []
    total_bits = 0
    for s, desc, nf in zip(keys, descending, nulls_first):
        kb = _key_bits(s)
        if kb is None:
            return None
        k, bits = kb
        if desc:
            k = ((1 << bits) - 1) - k
        nbit = 0
        if s.validity is not None:
            nbit = 1
            # null bit dominates the value: 0 sorts first
            nullv = 0 if nf else 1
            valuev = 1 - nullv
            k = k | torch.where(s.validity,
                                torch.full_like(k, valuev << bits),
                                torch.full_like(k, nullv << bits))
            # null rows: clear value bits so ordering among nulls is stable
            k = torch.where(s.validity, k,
                            torch.full_like(k, nullv << bits))
        total_bits += bits + nbit
        parts.append((k, bits + nbit))
    if total_bits > 63:
        return None
    packed = torch.zeros_like(parts[0][0])
    for k, bits in parts:                 # first key ends up in the MSBs
        packed = (packed << bits) | k
    return _stable_sort_perm_by(packed)


def argsort_multi(keys: Sequence[Series], descending: Sequence[bool],
                  nulls_first: Sequence[bool]) -> torch.Tensor:
    """Stable lexicographic argsort over multiple key columns."""
    n = len(keys[0])
    dev = keys[0].device
    if any(s.dtype.is_decimal() and s.children for s in keys):
        # wide decimals: expand into (hi signed, lo unsigned-ordered)
        # int64 sub-keys — the LSD loop below makes that lexicographic
        from . import decimal128 as d128
        ek, ed, en = [], [], []
        for s, de, nf in zip(keys, descending, nulls_first):
            if s.dtype.is_decimal() and s.children:
                lo, hi = d128.limbs(s)
                ek.append(Series(s.name, DataType.int64(), data=hi,
                                 validity=s.validity))
                ek.append(Series(s.name, DataType.int64(),
                                 data=d128.u_order_key(lo),
                                 validity=s.validity))
                ed.extend([de, de])
                en.extend([nf, nf])
            else:
                ek.append(s)
                ed.append(de)
                en.append(nf)
        keys, descending, nulls_first = ek, ed, en
    composed = _try_composed_argsort(keys, descending, nulls_first)
    if composed is not None:
        return composed
    perm = torch.arange(n, dtype=torch.int64, device=dev)
    # LSD over keys: sort by last key first
    for s, desc, nf in list(zip(keys, descending, nulls_first))[::-1]:
        if s.is_dict():
            # order-preserving code ranks: argsort the (tiny) vocab once
            vocab = s.children[0]
            vperm = argsort_multi([vocab], [False], [False])
            rank = torch.empty(len(vocab), dtype=torch.int64,
                               device=s.device)
            rank[vperm] = torch.arange(len(vocab), dtype=torch.int64,
                                       device=s.device)
            k = rank[s.data.to(torch.int64)]
            if desc:
                k = ~k
            kg = k[perm]
            perm = perm[_stable_sort_perm_by(kg)]
        elif s.dtype.kind in (TypeKind.STRING, TypeKind.BINARY):
            nchunks = max(1, (_max_strlen(s) + 7) // 8)
            for c in range(nchunks - 1, -1, -1):
                k = _string_chunk_key(s, c)
                if desc:
                    k = ~k
                kg = k[perm]
                perm = perm[_stable_sort_perm_by(kg)]
        else:
            k = _order_key_u64(s)
            if desc:
                k = ~k
            kg = k[perm]
            perm = perm[_stable_sort_perm_by(kg)]
        if s.validity is not None:
            # null-placement pass dominates value order (applied after):
            # encode so rows that should sort first get the smaller key
            null_key = 0 if nf else 1
            nk = torch.where(s.validity,
                             torch.full_like(perm, 1 - null_key),
                             torch.full_like(perm, null_key))
            perm = perm[_stable_sort_perm_by(nk[perm])]
    return perm


# ---------------------------------------------------------------------------
# hash partitioning (feeds RCCL all-to-all; ref: recordbatch ops/partition.rs)
# ---------------------------------------------------------------------------

def partition_by_hash(keys: Sequence[Series],
                      num_partitions: int) -> Tuple[torch.Tensor, torch.Tensor]:
    """Return (perm, counts): `perm` reorders rows grouped by partition id,
    `counts[p]` = number of rows in partition p."""
    h = hash_columns(keys)
    if h.is_cuda:
        u = h
        part = native_required().u64_mod(u, num_partitions)
    else:
        part = torch.from_numpy(
            (h.numpy().view(np.uint64) % np.uint64(num_partitions))
            .view(np.int64))
    counts = torch.bincount(part, minlength=num_partitions)
    perm = _stable_sort_perm_by(part)
    return perm, counts


def partition_random(n: int, num_partitions: int, seed: int,
                     device) -> Tuple[torch.Tensor, torch.Tensor]:
    g = torch.Generator(device="cpu").manual_seed(seed)
    part = torch.randint(0, num_partitions, (n,), generator=g,
                         dtype=torch.int64).to(device)
    counts = torch.bincount(part, minlength=num_partitions)
    perm = _stable_sort_perm_by(part)
    return perm, counts


def partition_by_value(parts: torch.Tensor,
                       num_partitions: int) -> Tuple[torch.Tensor, torch.Tensor]:
    counts = torch.bincount(parts.clamp(min=0), minlength=num_partitions)
    perm = _stable_sort_perm_by(parts)
    return perm, counts
