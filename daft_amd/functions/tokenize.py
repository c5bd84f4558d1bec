"""Tokenization (ref capability: /root/reference/src/daft-functions-tokenize
— tiktoken-style BPE encode/decode).  Uses the offline `tokenizers` wheel
when a local tokenizer file is given; ships a built-in byte-level fallback
("bytes") and whitespace hashing tokenizer ("simple") that need no
downloads."""
from __future__ import annotations

from typing import Optional

import torch

from ..schema import DataType
from ..series import Series

_TOKENIZERS = {}


def _load_tokenizer(name: str):
    if name in _TOKENIZERS:
        return _TOKENIZERS[name]
    tok = None
    if name.endswith(".json"):
        from tokenizers import Tokenizer
        tok = Tokenizer.from_file(name)
    _TOKENIZERS[name] = tok
    return tok


def tokenize_encode_series(s: Series, tokenizer: str = "simple") -> Series:
    if isinstance(tokenizer, str) and tokenizer.startswith("bpe:"):
        return bpe_encode_series(s, load_bpe(tokenizer))
    vals = s.cpu().to_pylist()
    out = []
    if tokenizer == "bytes":
        for v in vals:
            out.append(None if v is None else list(v.encode("utf-8")))
    elif tokenizer == "simple":
        # whitespace words -> stable 31-bit hashes (deterministic, offline)
        for v in vals:
            if v is None:
                out.append(None)
            else:
                toks = []
                for w in v.split():
                    h = 2166136261
                    for ch in w.encode():
                        h = ((h ^ ch) * 16777619) & 0x7FFFFFFF
                    toks.append(h)
                out.append(toks)
    else:
        tok = _load_tokenizer(tokenizer)
        if tok is None:
            raise ValueError(f"unknown tokenizer {tokenizer!r} (use 'simple',"
                             " 'bytes', or a local tokenizers .json path)")
        encs = tok.encode_batch([v if v is not None else "" for v in vals])
        out = [None if v is None else list(e.ids)
               for v, e in zip(vals, encs)]
    res = Series.from_pylist(s.name, out, DataType.list(DataType.int32()))
    return res.to(s.device) if s.is_gpu() else res


def tokenize_decode_series(s: Series, tokenizer: str = "simple") -> Series:
    if isinstance(tokenizer, str) and tokenizer.startswith("bpe:"):
        tok = load_bpe(tokenizer)
        vals = s.cpu().to_pylist()
        out = [None if v is None else tok.decode_py(v) for v in vals]
        res = Series.from_pylist(s.name, out, DataType.string())
        return res.to(s.device) if s.is_gpu() else res
    vals = s.cpu().to_pylist()
    if tokenizer == "bytes":
        out = [None if v is None else bytes(v).decode("utf-8", "replace")
               for v in vals]
    else:
        tok = _load_tokenizer(tokenizer)
        if tok is None:
            raise ValueError("decode requires 'bytes' or a tokenizers file")
        out = [None if v is None else tok.decode(list(v)) for v in vals]
    res = Series.from_pylist(s.name, out, DataType.string())
    return res.to(s.device) if s.is_gpu() else res


# ---------------------------------------------------------------------------
# byte-level BPE on GPU (ref: daft-functions-tokenize/src/bpe.rs; here the
# greedy merge loop runs one WAVEFRONT per row in csrc/bpe.hip)
# ---------------------------------------------------------------------------

def _bytes_to_unicode():
    """GPT-2's printable byte<->unicode table."""
    bs = (list(range(ord("!"), ord("~") + 1)) +
          list(range(0xA1, 0xAD)) + list(range(0xAE, 0x100)))
    cs = bs[:]
    n = 0
    for b in range(256):
        if b not in bs:
            bs.append(b)
            cs.append(256 + n)
            n += 1
    return dict(zip(bs, [chr(c) for c in cs]))


class BPETokenizer:
    """Byte-level BPE: vocab maps unicode-mapped byte strings to ids,
    merges are ranked (left, right) token-string pairs.  Greedy
    lowest-rank-first merging (the tiktoken/GPT-2 core loop, without
    regex pre-splitting)."""

    def __init__(self, vocab: dict, merges: list):
        b2u = _bytes_to_unicode()
        self.byte2id = torch.full((256,), -1, dtype=torch.int32)
        for b in range(256):
            tid = vocab.get(b2u[b])
            if tid is None:
                raise ValueError(f"vocab lacks byte symbol for 0x{b:02x}")
            self.byte2id[b] = tid
        self.vocab = vocab
        self.id2tok = {v: k for k, v in vocab.items()}
        self.ranks = {}          # (left_id, right_id) -> (rank, new_id)
        for rank, (l, r) in enumerate(merges):
            new = vocab.get(l + r)
            li, ri = vocab.get(l), vocab.get(r)
            if new is None or li is None or ri is None:
                continue
            self.ranks[(li, ri)] = (rank, new)
        self._tables = None

    @classmethod
    def from_tokenizers_json(cls, path: str) -> "BPETokenizer":
        import json
        with open(path) as f:
            doc = json.load(f)
        model = doc.get("model", doc)
        merges = [tuple(m.split(" ", 1)) if isinstance(m, str) else tuple(m)
                  for m in model["merges"]]
        return cls(model["vocab"], merges)

    def tables(self, device):
        """(byte2id, keys, vals) device tensors; open-addressing table."""
        if self._tables is None or str(self._tables[0].device) != str(device):
            n = max(2 * len(self.ranks), 8)
            size = 1 << (n - 1).bit_length()
            keys = torch.full((size,), -1, dtype=torch.int64)
            vals = torch.zeros(size, dtype=torch.int64)

            def h(key):
                x = key & 0xFFFFFFFFFFFFFFFF
                x ^= x >> 33
                x = (x * 0xff51afd7ed558ccd) & 0xFFFFFFFFFFFFFFFF
                x ^= x >> 29
                return x & (size - 1)
            for (a, b), (rank, new) in self.ranks.items():
                key = ((a & 0xFFFFFFFF) << 32) | (b & 0xFFFFFFFF)
                slot = h(key)
                while keys[slot] != -1:
                    slot = (slot + 1) & (size - 1)
                keys[slot] = key if key < (1 << 63) else key - (1 << 64)
                vals[slot] = ((rank << 32) | (new & 0xFFFFFFFF))
            self._tables = (self.byte2id.to(device), keys.to(device),
                            vals.to(device))
        return self._tables

    # CPU oracle / fallback (identical semantics to the kernel)
    def encode_py(self, data: bytes):
        ids = [int(self.byte2id[b]) for b in data]
        while len(ids) > 1:
            best = None
            for i in range(len(ids) - 1):
                r = self.ranks.get((ids[i], ids[i + 1]))
                if r is not None and (best is None or r[0] < best[0]):
                    best = (r[0], i, r[1])
            if best is None:
                break
            _rank, i, new = best
            ids[i:i + 2] = [new]
        return ids

    def decode_py(self, ids):
        b2u = _bytes_to_unicode()
        u2b = {v: k for k, v in b2u.items()}
        text = "".join(self.id2tok.get(int(i), "") for i in ids)
        return bytes(u2b[c] for c in text if c in u2b) \
            .decode("utf-8", "replace")


_BPE_CACHE: dict = {}


def load_bpe(spec) -> BPETokenizer:
    if isinstance(spec, BPETokenizer):
        return spec
    if spec not in _BPE_CACHE:
        _BPE_CACHE[spec] = BPETokenizer.from_tokenizers_json(
            spec[len("bpe:"):] if spec.startswith("bpe:") else spec)
    return _BPE_CACHE[spec]


def bpe_encode_series(s: Series, tok: BPETokenizer) -> Series:
    """utf8 -> list<int32> token ids; GPU rows run the wave-per-row HIP
    kernel, rows longer than the 4 KiB LDS cap fall back to the oracle."""
    from ..kernels import load_native
    if s.is_dict():
        s = s.dict_decode()
    nat = load_native()
    if s.is_gpu() and nat is not None:
        byte2id, keys, vals = tok.tables(s.device)
        out_ids, counts = nat.bpe_encode(s.offsets, s.data, byte2id,
                                         keys, vals)
        counts_c = counts.cpu()
        n = len(s)
        # assemble list<int32>: per-row slice [offset, offset+count)
        lens = counts_c.clamp(min=0).to(torch.int64)
        new_off = torch.zeros(n + 1, dtype=torch.int64)
        torch.cumsum(lens, 0, out=new_off[1:])
        total = int(new_off[-1])
        offs_c = s.offsets.cpu()
        idx_parts = []
        for i in range(n):
            c = int(lens[i])
            if c:
                idx_parts.append(torch.arange(
                    int(offs_c[i]), int(offs_c[i]) + c, dtype=torch.int64))
        idx = torch.cat(idx_parts) if idx_parts else \
            torch.zeros(0, dtype=torch.int64)
        flat = out_ids.cpu()[idx] if total else \
            torch.zeros(0, dtype=torch.int32)
        child = Series("item", DataType.int32(), data=flat)
        res = Series(s.name, DataType.list(DataType.int32()),
                     offsets=new_off, children=[child],
                     validity=s.validity.cpu() if s.validity is not None
                     else None)
        # host fallback for rows over the LDS cap
        over = (counts_c < 0).nonzero().reshape(-1)
        if int(over.numel()):
            vals_py = s.cpu().to_pylist()
            lists = res.to_pylist()
            for i in over.tolist():
                v = vals_py[i]
                lists[i] = None if v is None else \
                    tok.encode_py(v.encode("utf-8"))
            res = Series.from_pylist(s.name, lists,
                                     DataType.list(DataType.int32()))
        return res.to(s.device)
    vals = s.cpu().to_pylist()
    out = [None if v is None else tok.encode_py(v.encode("utf-8"))
           for v in vals]
    res = Series.from_pylist(s.name, out, DataType.list(DataType.int32()))
    return res.to(s.device) if s.is_gpu() else res
