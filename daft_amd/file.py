"""daft_amd.File — file-reference value for UDF columns (capability of
the reference's daft.File, /root/reference/daft/file/file.py:27-241:
a standard file interface over a path or in-memory bytes that UDFs can
open/read/seek without materializing the whole payload as a column)."""
from __future__ import annotations

import io
import mimetypes
import os
import tempfile
from typing import Optional, Union


class File:
    """Reference to a file: a local path or in-memory bytes."""

    def __init__(self, source: Union[str, bytes]):
        if isinstance(source, (bytes, bytearray)):
            self._bytes: Optional[bytes] = bytes(source)
            self._path: Optional[str] = None
        elif isinstance(source, str):
            self._bytes = None
            self._path = source
        else:
            raise TypeError(f"File() takes a path or bytes, got "
                            f"{type(source).__name__}")

    # -- identity ------------------------------------------------------
    @property
    def path(self) -> str:
        if self._path is None:
            raise ValueError("in-memory File has no path")
        return self._path

    @property
    def name(self) -> str:
        return os.path.basename(self._path) if self._path else "<memory>"

    def __str__(self):
        return f"File({self._path or '<memory>'})"

    __repr__ = __str__

    def __eq__(self, other):
        return isinstance(other, File) and \
            (self._path, self._bytes) == (other._path, other._bytes)

    def __hash__(self):
        return hash((self._path, self._bytes))

    # -- IO ------------------------------------------------------------
    def open(self, buffer_size: Optional[int] = None):
        """Open for reading; returns a seekable binary file object."""
        if self._bytes is not None:
            return io.BytesIO(self._bytes)
        return open(self._path, "rb",
                    buffering=buffer_size if buffer_size else -1)

    def read(self) -> bytes:
        if self._bytes is not None:
            return self._bytes
        with open(self._path, "rb") as f:
            return f.read()

    def size(self) -> int:
        if self._bytes is not None:
            return len(self._bytes)
        return os.path.getsize(self._path)

    def exists(self) -> bool:
        if self._bytes is not None:
            return True
        return os.path.exists(self._path)

    def readable(self) -> bool:
        return True

    def writable(self) -> bool:
        return False

    def seekable(self) -> bool:
        return True

    def isatty(self) -> bool:
        return False

    def mime_type(self) -> str:
        if self._path:
            guess, _ = mimetypes.guess_type(self._path)
            if guess:
                return guess
        data = self._bytes if self._bytes is not None else None
        if data is None and self._path and os.path.exists(self._path):
            with open(self._path, "rb") as f:
                data = f.read(16)
        if data:
            if data[:8] == b"\x89PNG\r\n\x1a\n":
                return "image/png"
            if data[:3] == b"\xff\xd8\xff":
                return "image/jpeg"
            if data[:4] == b"%PDF":
                return "application/pdf"
            if data[:4] == b"PK\x03\x04":
                return "application/zip"
        return "application/octet-stream"

    def is_image(self) -> bool:
        return self.mime_type().startswith("image/")

    def is_video(self) -> bool:
        return self.mime_type().startswith("video/")

    def is_audio(self) -> bool:
        return self.mime_type().startswith("audio/")

    def to_tempfile(self):
        """Copy contents into a NamedTemporaryFile (for libraries that
        need a real path)."""
        tmp = tempfile.NamedTemporaryFile(suffix=os.path.splitext(
            self._path or "")[1] or None)
        with self.open() as f:
            while True:
                chunk = f.read(1 << 20)
                if not chunk:
                    break
                tmp.write(chunk)
        tmp.flush()
        tmp.seek(0)
        return tmp


class DaftFileIO:
    """File-opening adapter over the object-store layer (ref:
    daft/file DaftFileIO): open(path) returns a binary file object for
    local or remote (s3/gs/az/hf/http) paths."""

    def __init__(self, io_config=None):
        self.io_config = io_config

    def open(self, path: str, mode: str = "rb"):
        if "w" in mode:
            raise NotImplementedError("DaftFileIO is read-only here")
        from .io.object_store import get_source, is_remote
        if is_remote(path):
            return get_source(path, self.io_config).open(path)
        return open(path, mode)
