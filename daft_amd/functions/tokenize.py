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
