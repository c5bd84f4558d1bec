"""Checkpoint / resume: pipeline-progress checkpointing that skips already-
processed source keys (ref: /root/reference/src/daft-checkpoint/ —
CheckpointStore trait store.rs:54, keys codec, RewriteCheckpointSource rule;
Python surface daft/checkpoint.py:25-50).

Usage:
    store = daft_amd.checkpoint.LocalCheckpointStore("/path/ckpt")
    cfg = daft_amd.checkpoint.CheckpointConfig(store, on="id")
    df = cfg.filter_processed(df)        # drops committed keys
    ... process / write ...
    cfg.commit(df)                        # records processed keys
"""
from __future__ import annotations

import json
import os
import threading
import uuid
from typing import Iterable, List, Optional, Set


class CheckpointStore:
    """Abstract processed-key store (ref: CheckpointStore trait)."""

    def committed_keys(self) -> Set:
        raise NotImplementedError

    def commit(self, keys: Iterable) -> None:
        raise NotImplementedError


class MemoryCheckpointStore(CheckpointStore):
    def __init__(self):
        self._keys: Set = set()
        self._lock = threading.Lock()

    def committed_keys(self) -> Set:
        with self._lock:
            return set(self._keys)

    def commit(self, keys: Iterable) -> None:
        with self._lock:
            self._keys.update(keys)


class LocalCheckpointStore(CheckpointStore):
    """Append-only JSONL key sets under a directory (the object-store layout
    of the reference's impls/s3.rs, on local paths — there is no object store
    in this environment)."""

    def __init__(self, directory: str):
        self.directory = directory
        os.makedirs(directory, exist_ok=True)

    def committed_keys(self) -> Set:
        out: Set = set()
        for f in sorted(os.listdir(self.directory)):
            if not f.endswith(".jsonl"):
                continue
            with open(os.path.join(self.directory, f)) as fh:
                for line in fh:
                    out.update(json.loads(line))
        return out

    def commit(self, keys: Iterable) -> None:
        keys = list(keys)
        if not keys:
            return
        path = os.path.join(self.directory,
                            f"keys_{uuid.uuid4().hex[:12]}.jsonl")
        tmp = path + ".tmp"
        with open(tmp, "w") as f:
            f.write(json.dumps(keys) + "\n")
        os.replace(tmp, path)  # atomic commit (IdempotentCommit analog)


class ObjectStoreCheckpointStore(CheckpointStore):
    """Processed-key sets as JSONL objects under an s3://-style prefix —
    the reference's object-store checkpoint (daft-checkpoint
    impls/s3.rs + keys_codec.rs) over daft_amd.io.object_store, with the
    same atomic per-commit-object idempotent layout."""

    def __init__(self, prefix: str, io_config=None):
        from .io.object_store import get_source
        self.prefix = prefix.rstrip("/") + "/"
        self.src = get_source(self.prefix, io_config)

    def committed_keys(self) -> Set:
        out: Set = set()
        for path, _sz in sorted(self.src.list_prefix(self.prefix)):
            if not path.endswith(".jsonl"):
                continue
            for line in self.src.get(path).decode().splitlines():
                if line.strip():
                    out.update(json.loads(line))
        return out

    def commit(self, keys: Iterable) -> None:
        keys = list(keys)
        if not keys:
            return
        path = f"{self.prefix}keys_{uuid.uuid4().hex[:12]}.jsonl"
        self.src.put(path, (json.dumps(keys) + "\n").encode())


class CheckpointConfig:
    def __init__(self, store: CheckpointStore, on: str):
        self.store = store
        self.on = on

    def filter_processed(self, df):
        """Drop rows whose key column is already committed (the
        RewriteCheckpointSource analog, applied at the DataFrame level)."""
        from . import from_pydict
        from .expressions import col
        committed = sorted(self.store.committed_keys())
        if not committed:
            return df
        keys_df = from_pydict({self.on: committed})
        return df.join(keys_df, on=self.on, how="anti")

    def commit(self, df) -> int:
        """Record the keys present in df as processed; returns count."""
        keys = df.select(self.on).distinct().to_pydict()[self.on]
        self.store.commit(keys)
        return len(keys)
