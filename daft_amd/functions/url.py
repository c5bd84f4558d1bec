"""url_download / url_upload (ref: /root/reference/src/daft-functions-uri/
src/download.rs:24-110, upload.rs:199 — async batched object fetch).  Local
files and file:// always work; http(s) uses a thread pool over requests."""
from __future__ import annotations

import concurrent.futures as _fut
import os
from typing import Optional

from ..schema import DataType
from ..series import Series

_MAX_WORKERS = 32


def _fetch_one(u: Optional[str], on_error: str):
    if u is None:
        return None
    try:
        if u.startswith("file://"):
            u = u[7:]
        from ..io.object_store import get_source, is_remote
        if is_remote(u):
            return get_source(u).get(u)
        with open(u, "rb") as f:
            return f.read()
    except Exception:
        if on_error == "raise":
            raise
        return None


def url_download_series(s: Series, on_error: str = "raise",
                        max_connections: int = _MAX_WORKERS) -> Series:
    urls = s.cpu().to_pylist()
    with _fut.ThreadPoolExecutor(max_workers=max_connections) as ex:
        out = list(ex.map(lambda u: _fetch_one(u, on_error), urls))
    res = Series.from_pylist(s.name, out, DataType.binary())
    return res.to(s.device) if s.is_gpu() else res


def url_upload_series(s: Series, paths: Series, location: str) -> Series:
    from ..io.object_store import get_source, is_remote
    data = s.cpu().to_pylist()
    names = paths.cpu().to_pylist()
    remote = is_remote(location)
    if not remote:
        os.makedirs(location, exist_ok=True)
        src = None
    else:
        src = get_source(location)
    out = []
    for d, n in zip(data, names):
        if d is None:
            out.append(None)
            continue
        payload = d if isinstance(d, bytes) else str(d).encode()
        if remote:
            p = location.rstrip("/") + "/" + str(n)
            src.put(p, payload)
        else:
            p = os.path.join(location, str(n))
            with open(p, "wb") as f:
                f.write(payload)
        out.append(p)
    res = Series.from_pylist(s.name, out, DataType.string())
    return res.to(s.device) if s.is_gpu() else res
