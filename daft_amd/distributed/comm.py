"""Collective exchange of RecordBatches over torch.distributed.

MI355X-native data plane (ref: the reference's daft-shuffles Arrow-Flight +
disk spill): partitions move directly between HBM buffers
with RCCL all-to-all over xGMI (backend "nccl" IS RCCL on ROCm); no disk
round trip.  The gloo backend (CPU test tier) uses object collectives.

A RecordBatch is flattened to an ordered list of tensors (data / validity /
offsets, recursively through children); each buffer position is exchanged
with one all_to_all_single, then batches are reassembled and concatenated.
"""
from __future__ import annotations

import io
import pickle
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist

from ..recordbatch import RecordBatch
from ..schema import Schema
from ..series import Series


def is_dist() -> bool:
    return dist.is_available() and dist.is_initialized()


def rank() -> int:
    return dist.get_rank() if is_dist() else 0


def world() -> int:
    return dist.get_world_size() if is_dist() else 1


def backend() -> str:
    return dist.get_backend() if is_dist() else "none"


# ---------------------------------------------------------------------------
# buffer flattening
# ---------------------------------------------------------------------------

def _flatten_series(s: Series, out: List[Optional[torch.Tensor]]):
    if s.pyobjs is not None:
        raise TypeError("python-object columns cannot be exchanged over RCCL")
    out.append(s.data)
    out.append(s.validity)
    out.append(s.offsets)
    for c in s.children:
        _flatten_series(c, out)


def _rebuild_series(proto: Series, bufs: List[Optional[torch.Tensor]],
                    pos: int) -> Tuple[Series, int]:
    data = bufs[pos]
    validity = bufs[pos + 1]
    offsets = bufs[pos + 2]
    pos += 3
    children = []
    for c in proto.children:
        ch, pos = _rebuild_series(c, bufs, pos)
        children.append(ch)
    length = None
    if offsets is not None:
        length = int(offsets.shape[0]) - 1
    elif data is not None:
        length = int(data.shape[0])
    s = Series(proto.name, proto.dtype, data=data, validity=validity,
               offsets=offsets, children=children, length=length)
    if length is None:
        # fixed-size-list style: length from children
        s._length = Series(proto.name, proto.dtype, children=children)._length
    return s, pos


def flatten_batch(rb: RecordBatch) -> List[Optional[torch.Tensor]]:
    bufs: List[Optional[torch.Tensor]] = []
    for c in rb.columns:
        _flatten_series(c, bufs)
    return bufs


def rebuild_batch(proto: RecordBatch,
                  bufs: List[Optional[torch.Tensor]]) -> RecordBatch:
    cols = []
    pos = 0
    for c in proto.columns:
        s, pos = _rebuild_series(c, bufs, pos)
        cols.append(s)
    n = len(cols[0]) if cols else 0
    return RecordBatch(cols, num_rows=n)


# ---------------------------------------------------------------------------
# exchanges
# ---------------------------------------------------------------------------

_ALIGN = 16  # byte alignment of packed buffers: any dtype views cleanly


def _align(n: int) -> int:
    return (n + _ALIGN - 1) & ~(_ALIGN - 1)


_TOK2DT = {"b1": torch.bool, "i1": torch.int8, "u1": torch.uint8,
           "i2": torch.int16, "u2": torch.uint16,
           "i4": torch.int32, "u4": torch.uint32,
           "i8": torch.int64, "u8": torch.uint64,
           "f4": torch.float32, "f8": torch.float64}


def _dt_tok(dt: torch.dtype) -> str:
    return _DT_TOKENS[dt]


def _structure_skel(rb: RecordBatch):
    def srec(s: Series):
        return (repr(s.dtype), s.is_dict(),
                tuple(srec(c) for c in s.children))
    return tuple(srec(c) for c in rb.columns)


def _observed_dtypes(flat_lists) -> List[Optional[str]]:
    """Per buffer position, the dtype token of any non-None tensor."""
    nbuf = len(flat_lists[0])
    out: List[Optional[str]] = [None] * nbuf
    for flat in flat_lists:
        for b, t in enumerate(flat):
            if t is not None and out[b] is None:
                out[b] = _dt_tok(t.dtype)
    return out


def exchange_batches(parts: List[RecordBatch],
                     hbm_budget: Optional[int] = None) -> RecordBatch:
    """All-to-all: parts[p] goes to rank p; returns concat of received.

    Every buffer of every partition is packed into ONE contiguous byte
    buffer per destination and moved with a single all_to_all_single (RCCL
    over xGMI on GPU; the identical tensor protocol runs on gloo for the
    CPU test tier — ref: daft-shuffles shuffle_cache.rs replaced by direct
    HBM exchange).  Receivers view slices back out
    zero-copy (16-byte alignment keeps every dtype viewable).

    With `hbm_budget` (bytes), a receive total that would exceed it runs
    the exchange in row chunks, staging each received chunk to host — the
    spill tier standing in for the reference's flight_shuffle_dirs disk
    spill.  The result batch then lives on the host; callers stream it
    back per morsel.
    """
    w = world()
    assert len(parts) == w
    if w == 1:
        return parts[0]
    if any(c.pyobjs is not None for p in parts for c in p.columns):
        if backend() == "nccl":
            raise TypeError(
                "python-object columns cannot be exchanged over RCCL")
        return RecordBatch.concat(_object_a2a(parts))

    # The buffer layout must agree across ranks (dictionary-encoded columns
    # add a vocab child).  Structures only diverge through data-dependent
    # decode fallbacks; agree on a canonical structure first (tiny
    # control-plane collective), decoding dicts everywhere on mismatch.
    my_skel = _structure_skel(parts[0])
    my_dts = _observed_dtypes([flatten_batch(p) for p in parts])
    ctrl: List[object] = [None] * w
    dist.all_gather_object(ctrl, (my_skel, my_dts))
    if any(sk != my_skel for sk, _ in ctrl):
        parts = [RecordBatch(
            [c.dict_decode() if c.is_dict() else c for c in p.columns],
            len(p)) for p in parts]
        my_skel = _structure_skel(parts[0])
        my_dts = _observed_dtypes([flatten_batch(p) for p in parts])
        ctrl = [None] * w
        dist.all_gather_object(ctrl, (my_skel, my_dts))
    # merge per-position dtypes across ranks (a validity bitmap may exist
    # on some ranks only)
    nbuf = len(my_dts)
    if any(len(rd) != nbuf for _, rd in ctrl):
        # buffer counts disagree even though skeletons matched: the ranks
        # are in DIFFERENT exchanges — an SPMD plan-divergence bug.  Fail
        # loudly instead of corrupting data.
        raise RuntimeError(
            f"exchange control mismatch (rank {rank()}): buffer counts "
            f"{[len(rd) for _, rd in ctrl]} — ranks are executing "
            f"different plans (SPMD divergence)")
    dts: List[Optional[str]] = list(my_dts)
    for _, rd in ctrl:
        for b in range(nbuf):
            if dts[b] is None:
                dts[b] = rd[b]
            elif rd[b] is not None and rd[b] != dts[b]:
                raise RuntimeError(
                    f"exchange dtype disagreement at buffer {b}: "
                    f"{dts[b]} vs {rd[b]}")

    proto = parts[0]
    flat = [flatten_batch(p) for p in parts]
    dev = proto.device
    wire_dev = dev if backend() == "nccl" else torch.device("cpu")

    # exchange element counts (one int64 collective)
    sizes = torch.full((w, nbuf), -1, dtype=torch.int64)
    for p in range(w):
        for b, t in enumerate(flat[p]):
            if t is not None:
                sizes[p, b] = t.numel()
    sizes = sizes.to(wire_dev)
    recv_sizes = torch.empty_like(sizes)
    dist.all_to_all_single(recv_sizes, sizes.contiguous())
    recv_cpu = recv_sizes.cpu()
    sizes_cpu = sizes.cpu()

    def packed_total(counts_row) -> int:
        off = 0
        for b in range(nbuf):
            c = int(counts_row[b])
            if c >= 0 and dts[b] is not None:
                isz = _TOK2DT[dts[b]].itemsize
                off = _align(off) + c * isz
        return _align(off)

    recv_total = sum(packed_total(recv_cpu[p]) for p in range(w))
    send_total = sum(packed_total(sizes_cpu[p]) for p in range(w))
    if hbm_budget is not None:
        # the spill decision MUST be collective: totals are rank-local
        # (skewed exchanges) and divergent control flow deadlocks the
        # collective schedule
        ws = torch.tensor([recv_total + send_total], dtype=torch.int64,
                          device=wire_dev)
        dist.all_reduce(ws, op=dist.ReduceOp.MAX)
        max_ws = int(ws.item())
        if max_ws > hbm_budget:
            return _chunked_exchange(parts, hbm_budget, max_ws)

    # pack: one contiguous byte buffer per destination
    in_splits = [packed_total(sizes_cpu[p]) for p in range(w)]
    sendbuf = torch.zeros(sum(in_splits), dtype=torch.uint8, device=wire_dev)
    base = 0
    for p in range(w):
        off = 0
        for b, t in enumerate(flat[p]):
            if t is None:
                continue
            isz = t.dtype.itemsize
            off = _align(off)
            nb = t.numel() * isz
            if nb:
                src = t.contiguous().view(-1).view(torch.uint8)
                if src.device != wire_dev:
                    src = src.to(wire_dev)
                sendbuf[base + off:base + off + nb] = src
            off += nb
        base += in_splits[p]

    out_splits = [packed_total(recv_cpu[p]) for p in range(w)]
    recvbuf = torch.empty(sum(out_splits), dtype=torch.uint8,
                          device=wire_dev)
    dist.all_to_all_single(recvbuf, sendbuf, out_splits, in_splits)
    if wire_dev != dev:
        recvbuf = recvbuf.to(dev)

    # unpack: zero-copy views into recvbuf
    received = []
    base = 0
    for p in range(w):
        bufs_p: List[Optional[torch.Tensor]] = []
        off = 0
        for b in range(nbuf):
            c = int(recv_cpu[p, b])
            if c < 0 or dts[b] is None:
                bufs_p.append(None)
                continue
            dt = _TOK2DT[dts[b]]
            isz = dt.itemsize
            off = _align(off)
            nb = c * isz
            bufs_p.append(recvbuf[base + off:base + off + nb].view(dt))
            off += nb
        received.append(rebuild_batch(proto, bufs_p))
        base += out_splits[p]
    return RecordBatch.concat(received)


def _chunked_exchange(parts: List[RecordBatch], hbm_budget: int,
                      max_ws: int) -> RecordBatch:
    """Spill tier: exchange in row chunks, staging each received chunk to
    host memory so peak HBM stays within budget (the xGMI analog of the
    reference's flight_shuffle_dirs disk spill, shuffle_cache.rs:47-90).
    `max_ws` is the GLOBAL max working set (already all-reduced), so every
    rank derives the same chunk count — a rank-local value here would
    desynchronize the collective schedule."""
    nchunks = max(2, -(-max_ws * 2 // max(hbm_budget, 1)))
    nchunks = min(nchunks, 1024)
    host_parts: List[RecordBatch] = []
    for k in range(nchunks):
        chunk = []
        for p in parts:
            n = len(p)
            lo = (n * k) // nchunks
            hi = (n * (k + 1)) // nchunks
            chunk.append(p.slice(lo, hi))
        got = exchange_batches(chunk)  # no budget: bounded by chunking
        host_parts.append(got.cpu())
    return RecordBatch.concat(host_parts)


def _object_a2a(parts: List[RecordBatch]) -> List[RecordBatch]:
    """gloo/test fallback: object all-to-all via all_gather_object."""
    w = world()
    payload = [_pickle_batch(p) for p in parts]
    gathered: List[list] = [None] * w  # type: ignore
    dist.all_gather_object(gathered, payload)
    mine = [_unpickle_batch(gathered[src][rank()]) for src in range(w)]
    return mine


_DT_TOKENS = {
    torch.bool: "b1", torch.int8: "i1", torch.int16: "i2",
    torch.int32: "i4", torch.int64: "i8", torch.uint8: "u1",
    torch.uint16: "u2", torch.uint32: "u4", torch.uint64: "u8",
    torch.float32: "f4", torch.float64: "f8",
}
_SIGNED_VIEW = {torch.uint16: torch.int16, torch.uint32: torch.int32,
                torch.uint64: torch.int64}


def _t2blob(t: Optional[torch.Tensor]):
    if t is None:
        return None
    tok = _DT_TOKENS[t.dtype]
    v = t
    if t.dtype in _SIGNED_VIEW:
        v = t.view(_SIGNED_VIEW[t.dtype])
    import numpy as np
    return (tok, v.contiguous().numpy().tobytes())


def _blob2t(blob) -> Optional[torch.Tensor]:
    if blob is None:
        return None
    tok, raw = blob
    import numpy as np
    np_dt = {"b1": np.bool_, "i1": np.int8, "i2": np.int16, "i4": np.int32,
             "i8": np.int64, "u1": np.uint8, "u2": np.uint16,
             "u4": np.uint32, "u8": np.uint64, "f4": np.float32,
             "f8": np.float64}[tok]
    arr = np.frombuffer(bytearray(raw), dtype=np_dt)
    if tok in ("u2", "u4", "u8"):
        signed = arr.view({"u2": np.int16, "u4": np.int32,
                           "u8": np.int64}[tok])
        t = torch.from_numpy(signed.copy())
        return t.view({"u2": torch.uint16, "u4": torch.uint32,
                       "u8": torch.uint64}[tok])
    return torch.from_numpy(arr.copy())


def _skeleton(s: Series):
    return (s.name, s.dtype, [_skeleton(c) for c in s.children])


def _proto_from_skeleton(sk) -> Series:
    name, dtype, children = sk
    return Series(name, dtype, children=[_proto_from_skeleton(c)
                                         for c in children], length=0)


def _pickle_batch(rb: RecordBatch) -> bytes:
    cpu = rb.cpu()
    cols = []
    for c in cpu.columns:
        if c.pyobjs is not None:
            cols.append(("py", (c.name, c.dtype), list(c.pyobjs),
                         _t2blob(c.validity)))
        else:
            bufs: List[Optional[torch.Tensor]] = []
            _flatten_series(c, bufs)
            cols.append(("t", _skeleton(c), [_t2blob(t) for t in bufs]))
    return pickle.dumps((cols, len(cpu)))


def _unpickle_batch(blob: bytes) -> RecordBatch:
    cols, n = pickle.loads(blob)
    series = []
    for rec in cols:
        if rec[0] == "py":
            (_, (name, dtype), objs, vblob) = rec
            series.append(Series(name, dtype, pyobjs=objs,
                                 validity=_blob2t(vblob)))
        else:
            (_, sk, blobs) = rec
            bufs = [_blob2t(b) for b in blobs]
            s, _pos = _rebuild_series(_proto_from_skeleton(sk), bufs, 0)
            series.append(s)
    return RecordBatch(series, num_rows=n)


def allgather_batch(rb: RecordBatch, device=None) -> RecordBatch:
    """Every rank receives the concatenation of all ranks' batches (rank
    order preserved) — the RCCL broadcast/gather analog of the reference's
    GatherSink + broadcast join build replication.  Implemented as a
    replicating all-to-all: every rank sends its batch to every peer
    simultaneously, which drives all 7 xGMI links at once (a ring
    broadcast would be per-link bound)."""
    w = world()
    if w == 1:
        return rb
    if any(c.pyobjs is not None for c in rb.columns):
        gathered: List[bytes] = [None] * w  # type: ignore
        dist.all_gather_object(gathered, _pickle_batch(rb))
        return RecordBatch.concat([_unpickle_batch(g) for g in gathered])
    return exchange_batches([rb for _ in range(w)])


def gather_pydict(d: dict) -> dict:
    """Allgather small result dicts (rank-ordered merge)."""
    if not is_dist():
        return d
    w = world()
    gathered: List[dict] = [None] * w  # type: ignore
    dist.all_gather_object(gathered, d)
    out = {k: [] for k in gathered[0]} if gathered[0] is not None else {}
    for g in gathered:
        for k, v in g.items():
            out.setdefault(k, []).extend(v)
    return out
