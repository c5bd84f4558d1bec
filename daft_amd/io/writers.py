"""Write path: target-file-size batching + hive-partitioned fanout (ref:
/root/reference/src/daft-writers/src/lib.rs:136-198 and sinks/write.rs)."""
from __future__ import annotations

import os
import uuid
from typing import Iterator, List

from ..recordbatch import RecordBatch


def write_batches(batches: Iterator[RecordBatch], file_format: str,
                  root_dir: str, write_mode: str, partition_cols,
                  options: dict, ectx) -> List[str]:
    from .object_store import is_remote
    if not is_remote(root_dir):
        os.makedirs(root_dir, exist_ok=True)
    if write_mode == "overwrite" and not is_remote(root_dir):
        for f in os.listdir(root_dir):
            p = os.path.join(root_dir, f)
            if os.path.isfile(p) and f.endswith((".parquet", ".csv",
                                                 ".json", ".jsonl")):
                os.remove(p)
    paths: List[str] = []
    target = options.get("target_filesize", 512 * 1024 * 1024)
    pending: List[RecordBatch] = []
    pending_bytes = 0

    def flush():
        nonlocal pending, pending_bytes
        if not pending:
            return
        rb = RecordBatch.concat(pending) if len(pending) > 1 else pending[0]
        pending, pending_bytes = [], 0
        if partition_cols:
            _write_partitioned(rb, file_format, root_dir, partition_cols,
                               options, paths)
        else:
            paths.append(_write_one(rb, file_format, root_dir, options))

    for rb in batches:
        pending.append(rb)
        pending_bytes += rb.size_bytes()
        if pending_bytes >= target:
            flush()
    flush()
    return paths


def _write_partitioned(rb: RecordBatch, file_format, root_dir,
                       partition_cols, options, paths: List[str]):
    key_series = [e.evaluate(rb) for e in partition_cols]
    from ..kernels import rowops
    gids, reps = rowops.groupby(key_series)
    import torch
    for g in range(int(reps.shape[0])):
        mask = gids == g
        from ..series import Series
        from ..schema import DataType
        part = rb.filter(Series("m", DataType.bool(), data=mask))
        kv = []
        for s in key_series:
            v = s.take(reps[g:g + 1]).cpu().to_pylist()[0]
            kv.append(f"{s.name}={v}")
        sub = os.path.join(root_dir, *kv)
        os.makedirs(sub, exist_ok=True)
        paths.append(_write_one(part, file_format, sub, options))


def _join(dir_: str, name: str) -> str:
    from .object_store import is_remote
    if is_remote(dir_):
        return dir_.rstrip("/") + "/" + name
    return os.path.join(dir_, name)


def _write_one(rb: RecordBatch, file_format: str, dir_: str,
               options: dict) -> str:
    from .object_store import get_source, is_remote
    if is_remote(dir_):
        # encode to a buffer, then one put (multipart for large objects)
        import io as _io
        name = uuid.uuid4().hex[:16]
        tbl = rb.to_arrow()
        buf = _io.BytesIO()
        if file_format == "parquet":
            import pyarrow.parquet as pq
            pq.write_table(tbl, buf,
                           compression=options.get("compression", "snappy"),
                           row_group_size=options.get("row_group_size",
                                                      1 << 20))
            path = _join(dir_, f"{name}.parquet")
        elif file_format == "csv":
            import pyarrow.csv as pacsv
            pacsv.write_csv(tbl, buf)
            path = _join(dir_, f"{name}.csv")
        else:
            raise ValueError(
                f"remote write format {file_format} not supported")
        get_source(path, options.get("io_config")).put(path,
                                                       buf.getvalue())
        return path
    name = uuid.uuid4().hex[:16]
    tbl = rb.to_arrow()
    if file_format == "parquet":
        import pyarrow.parquet as pq
        path = os.path.join(dir_, f"{name}.parquet")
        pq.write_table(tbl, path,
                       compression=options.get("compression", "snappy"),
                       row_group_size=options.get("row_group_size",
                                                  1 << 20))
    elif file_format == "csv":
        import pyarrow.csv as pacsv
        path = os.path.join(dir_, f"{name}.csv")
        pacsv.write_csv(tbl, path)
    elif file_format == "ipc":
        import pyarrow as pa
        path = os.path.join(dir_, f"{name}.arrow")
        with pa.ipc.new_file(path, tbl.schema) as w:
            w.write_table(tbl)
    elif file_format in ("json", "jsonl"):
        path = os.path.join(dir_, f"{name}.jsonl")
        import json
        with open(path, "w") as fh:
            d = rb.to_pydict()
            names = list(d.keys())
            for i in range(len(rb)):
                fh.write(json.dumps({n: _jsonable(d[n][i]) for n in names},
                                    default=str))
                fh.write("\n")
    else:
        raise ValueError(f"unknown write format {file_format}")
    return path


def _jsonable(v):
    import numpy as np
    if isinstance(v, (np.integer,)):
        return int(v)
    if isinstance(v, (np.floating,)):
        return float(v)
    if isinstance(v, np.ndarray):
        return v.tolist()
    return v
