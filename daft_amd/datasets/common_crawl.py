"""Common Crawl loader (ref: /root/reference/daft/datasets/common_crawl.py
:17-76 — resolve the crawl's `<file_type>.paths.gz` manifest, optionally
filter by segment, limit to num_files, then read the WARC/WET/WAT files).

Offline-first: `data_root` names a local mirror of the
`crawl-data/<crawl>/...` layout (an s3://commoncrawl/-style tree on
disk); with no egress in this environment remote sources raise."""
from __future__ import annotations

import gzip
import os
from typing import Optional


_CONTENT_TO_FILETYPE = {"raw": "warc", "warc": "warc",
                        "text": "wet", "wet": "wet",
                        "metadata": "wat", "wat": "wat"}


def _manifest_paths(data_root: str, crawl: str, file_type: str,
                    segment: Optional[str], num_files: Optional[int]):
    man = os.path.join(data_root, "crawl-data", crawl,
                       f"{file_type}.paths.gz")
    if not os.path.exists(man):
        raise FileNotFoundError(
            f"Could not find the crawl manifest {man!r} (data_root must "
            f"mirror the commoncrawl bucket layout)")
    opener = gzip.open if man.endswith(".gz") else open
    with opener(man, "rt") as f:
        rels = [ln.strip() for ln in f if ln.strip()]
    if segment is not None:
        rels = [r for r in rels if segment in r]
    if num_files is not None:
        rels = rels[:num_files]
    return [os.path.join(data_root, r) for r in rels]


def common_crawl(crawl: str, segment: Optional[str] = None,
                 content: str = "raw", num_files: Optional[int] = None,
                 io_config=None, *, data_root: Optional[str] = None,
                 source: Optional[str] = None):
    """Load Common Crawl data as a DataFrame (WARC schema: see
    daft_amd.io.readers.warc_schema)."""
    from .. import read_warc
    if content not in _CONTENT_TO_FILETYPE:
        raise ValueError(f"unknown content {content!r}; expected one of "
                         f"{sorted(_CONTENT_TO_FILETYPE)}")
    if data_root is None:
        raise ValueError(
            "this build has no network egress: pass data_root= pointing "
            "at a local mirror of the commoncrawl bucket")
    ft = _CONTENT_TO_FILETYPE[content]
    paths = _manifest_paths(data_root, crawl, ft, segment, num_files)
    if not paths:
        raise FileNotFoundError(
            f"no {ft} files in crawl {crawl!r} match segment={segment!r}")
    return read_warc(paths, io_config=io_config)
