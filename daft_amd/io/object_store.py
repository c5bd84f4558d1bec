"""Object-store IO: S3-compatible / HTTP / local sources with ranged
reads, multipart upload, prefix listing, `**` globbing and retry with
exponential backoff.

MI355X-native counterpart of the reference's async Rust IO layer
(/root/reference/src/daft-io/src/object_io.rs:289-320 ObjectSource trait,
s3_like.rs, http.rs, object_store_glob.rs, retry.rs;
/root/reference/src/common/io-config/src/s3.rs:20-44 S3Config): host IO
feeds the GPU decode path, so it stays Python (requests) with hand-rolled
SigV4 — no SDK dependencies exist in the image.
"""
from __future__ import annotations

import datetime as _dt
import fnmatch
import hashlib
import hmac
import io as _io
import os
import random
import re
import time
import urllib.parse
from dataclasses import dataclass, field
from typing import Iterator, List, Optional, Tuple


# ---------------------------------------------------------------------------
# config (names mirror the reference's io-config)
# ---------------------------------------------------------------------------

@dataclass
class S3Config:
    region_name: str = "us-east-1"
    endpoint_url: Optional[str] = None
    key_id: Optional[str] = None
    access_key: Optional[str] = None
    session_token: Optional[str] = None
    anonymous: bool = False
    max_connections_per_io_thread: int = 8
    num_tries: int = 5
    retry_initial_backoff_ms: int = 100
    connect_timeout_ms: int = 10_000
    read_timeout_ms: int = 30_000
    verify_ssl: bool = True
    force_virtual_addressing: bool = False

    def resolved(self) -> "S3Config":
        c = S3Config(**self.__dict__)
        c.key_id = c.key_id or os.environ.get("AWS_ACCESS_KEY_ID")
        c.access_key = c.access_key or os.environ.get("AWS_SECRET_ACCESS_KEY")
        c.session_token = c.session_token or os.environ.get("AWS_SESSION_TOKEN")
        c.endpoint_url = c.endpoint_url or os.environ.get("AWS_ENDPOINT_URL")
        return c


@dataclass
class HTTPConfig:
    num_tries: int = 4
    retry_initial_backoff_ms: int = 100
    read_timeout_ms: int = 30_000


@dataclass
class IOConfig:
    s3: S3Config = field(default_factory=S3Config)
    http: HTTPConfig = field(default_factory=HTTPConfig)


# ---------------------------------------------------------------------------
# retry
# ---------------------------------------------------------------------------

class ObjectStoreError(IOError):
    pass


class NotFoundError(ObjectStoreError):
    pass


def _with_retry(fn, num_tries: int, backoff_ms: int, what: str):
    """Exponential backoff + full jitter on transient failures
    (ref: daft-io/src/retry.rs)."""
    last = None
    for attempt in range(max(1, num_tries)):
        try:
            return fn()
        except NotFoundError:
            raise
        except Exception as e:  # connection errors, 5xx, 429
            last = e
            if attempt + 1 >= num_tries:
                break
            delay = backoff_ms * (2 ** attempt) / 1000.0
            time.sleep(random.uniform(0, delay))
    raise ObjectStoreError(f"{what} failed after {num_tries} tries: {last}")


# ---------------------------------------------------------------------------
# SigV4
# ---------------------------------------------------------------------------

def _hmac(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def sigv4_headers(method: str, url: str, region: str, key_id: str,
                  secret: str, payload: bytes,
                  session_token: Optional[str] = None,
                  service: str = "s3") -> dict:
    """AWS Signature Version 4 for S3-style requests."""
    parsed = urllib.parse.urlsplit(url)
    host = parsed.netloc
    canonical_uri = urllib.parse.quote(parsed.path or "/", safe="/~-._")
    qs_items = urllib.parse.parse_qsl(parsed.query, keep_blank_values=True)
    qs_items.sort()
    canonical_qs = "&".join(
        f"{urllib.parse.quote(k, safe='~-._')}="
        f"{urllib.parse.quote(v, safe='~-._')}" for k, v in qs_items)
    now = _dt.datetime.now(_dt.timezone.utc)
    amzdate = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")
    payload_hash = hashlib.sha256(payload).hexdigest()
    headers = {"host": host, "x-amz-content-sha256": payload_hash,
               "x-amz-date": amzdate}
    if session_token:
        headers["x-amz-security-token"] = session_token
    signed = ";".join(sorted(headers))
    canonical_headers = "".join(f"{k}:{headers[k]}\n" for k in sorted(headers))
    creq = "\n".join([method, canonical_uri, canonical_qs,
                      canonical_headers, signed, payload_hash])
    scope = f"{datestamp}/{region}/{service}/aws4_request"
    sts = "\n".join(["AWS4-HMAC-SHA256", amzdate, scope,
                     hashlib.sha256(creq.encode()).hexdigest()])
    k = _hmac(("AWS4" + secret).encode(), datestamp)
    k = _hmac(k, region)
    k = _hmac(k, service)
    k = _hmac(k, "aws4_request")
    sig = hmac.new(k, sts.encode(), hashlib.sha256).hexdigest()
    out = dict(headers)
    out["Authorization"] = (
        f"AWS4-HMAC-SHA256 Credential={key_id}/{scope}, "
        f"SignedHeaders={signed}, Signature={sig}")
    del out["host"]  # requests sets it
    return out


# ---------------------------------------------------------------------------
# sources
# ---------------------------------------------------------------------------

class ObjectSource:
    """ref: daft-io ObjectSource trait (object_io.rs:289-320)."""

    def get(self, path: str,
            range_: Optional[Tuple[int, int]] = None) -> bytes:
        raise NotImplementedError

    def get_size(self, path: str) -> int:
        raise NotImplementedError

    def put(self, path: str, data: bytes) -> None:
        raise NotImplementedError

    def list_prefix(self, path_prefix: str) -> List[Tuple[str, int]]:
        """[(full path, size)] under the prefix (recursive)."""
        raise NotImplementedError

    def glob(self, pattern: str) -> List[str]:
        prefix = _static_prefix(pattern)
        rx = _glob_to_regex(pattern)
        return sorted(p for p, _sz in self.list_prefix(prefix)
                      if rx.match(p))

    def open(self, path: str) -> _io.BytesIO:
        return _io.BytesIO(self.get(path))


def _static_prefix(pattern: str) -> str:
    """Longest prefix before any glob metacharacter."""
    m = re.search(r"[\*\?\[]", pattern)
    head = pattern if m is None else pattern[:m.start()]
    return head[:head.rfind("/") + 1] if "/" in head else head


def _glob_to_regex(pattern: str) -> "re.Pattern":
    """fnmatch-style with `**` crossing directory boundaries
    (ref: daft-io/src/object_store_glob.rs)."""
    out = []
    i = 0
    while i < len(pattern):
        c = pattern[i]
        if c == "*":
            if pattern[i:i + 2] == "**":
                out.append(".*")
                i += 2
                if i < len(pattern) and pattern[i] == "/":
                    i += 1
                continue
            out.append("[^/]*")
        elif c == "?":
            out.append("[^/]")
        elif c in ".^$+{}()|[]\\":
            out.append("\\" + c)
        else:
            out.append(c)
        i += 1
    return re.compile("".join(out) + "$")


class LocalSource(ObjectSource):
    def get(self, path, range_=None):
        with open(path, "rb") as f:
            if range_ is None:
                return f.read()
            f.seek(range_[0])
            return f.read(range_[1] - range_[0])

    def get_size(self, path):
        return os.path.getsize(path)

    def put(self, path, data):
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with open(path, "wb") as f:
            f.write(data)

    def list_prefix(self, path_prefix):
        base = path_prefix if os.path.isdir(path_prefix) else \
            os.path.dirname(path_prefix)
        out = []
        for root, _dirs, files in os.walk(base or "."):
            for fn in files:
                p = os.path.join(root, fn)
                out.append((p, os.path.getsize(p)))
        return out


class HTTPSource(ObjectSource):
    def __init__(self, config: Optional[HTTPConfig] = None):
        self.cfg = config or HTTPConfig()
        import requests
        self._sess = requests.Session()

    def get(self, path, range_=None):
        def go():
            headers = {}
            if range_ is not None:
                headers["Range"] = f"bytes={range_[0]}-{range_[1] - 1}"
            r = self._sess.get(path, headers=headers,
                               timeout=self.cfg.read_timeout_ms / 1000)
            if r.status_code == 404:
                raise NotFoundError(path)
            if r.status_code >= 400:
                raise ObjectStoreError(f"HTTP {r.status_code} for {path}")
            return r.content
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms, f"GET {path}")

    def get_size(self, path):
        def go():
            r = self._sess.head(path,
                                timeout=self.cfg.read_timeout_ms / 1000)
            if r.status_code == 404:
                raise NotFoundError(path)
            if r.status_code >= 400 or "Content-Length" not in r.headers:
                raise ObjectStoreError(f"HEAD {r.status_code} for {path}")
            return int(r.headers["Content-Length"])
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms, f"HEAD {path}")


class S3Source(ObjectSource):
    """S3-compatible store over the REST API (AWS, minio, any mock
    implementing GET/PUT/HEAD/ListObjectsV2).  Ref: daft-io s3_like.rs."""

    MULTIPART_CHUNK = 8 * 1024 * 1024

    def __init__(self, config: Optional[S3Config] = None):
        self.cfg = (config or S3Config()).resolved()
        import requests
        self._sess = requests.Session()

    # s3://bucket/key -> (endpoint url, bucket, key)
    def _url(self, path: str, query: str = "") -> Tuple[str, str, str]:
        parsed = urllib.parse.urlsplit(path)
        bucket, key = parsed.netloc, parsed.path.lstrip("/")
        if self.cfg.endpoint_url:
            base = self.cfg.endpoint_url.rstrip("/")
            url = f"{base}/{bucket}/{urllib.parse.quote(key)}"
        else:
            url = (f"https://{bucket}.s3.{self.cfg.region_name}"
                   f".amazonaws.com/{urllib.parse.quote(key)}")
        if query:
            url += "?" + query
        return url, bucket, key

    def _headers(self, method: str, url: str, payload: bytes = b"") -> dict:
        if self.cfg.anonymous or not self.cfg.key_id:
            return {}
        return sigv4_headers(method, url, self.cfg.region_name,
                             self.cfg.key_id, self.cfg.access_key or "",
                             payload, self.cfg.session_token)

    def _request(self, method: str, url: str, what: str, payload: bytes = b"",
                 extra_headers: Optional[dict] = None, ok=(200,)):
        def go():
            headers = self._headers(method, url, payload)
            if extra_headers:
                headers.update(extra_headers)
            r = self._sess.request(
                method, url, data=payload if method in ("PUT", "POST")
                else None, headers=headers,
                timeout=(self.cfg.connect_timeout_ms / 1000,
                         self.cfg.read_timeout_ms / 1000),
                verify=self.cfg.verify_ssl)
            if r.status_code == 404:
                raise NotFoundError(url)
            if r.status_code not in ok and r.status_code != 206:
                raise ObjectStoreError(
                    f"S3 {r.status_code} for {what}: {r.text[:200]}")
            return r
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms, what)

    def get(self, path, range_=None):
        url, _b, _k = self._url(path)
        extra = {}
        if range_ is not None:
            extra["Range"] = f"bytes={range_[0]}-{range_[1] - 1}"
        return self._request("GET", url, f"GET {path}",
                             extra_headers=extra).content

    def get_size(self, path):
        url, _b, _k = self._url(path)
        r = self._request("HEAD", url, f"HEAD {path}")
        return int(r.headers["Content-Length"])

    def put(self, path, data: bytes):
        if len(data) > self.MULTIPART_CHUNK * 2:
            return self._put_multipart(path, data)
        url, _b, _k = self._url(path)
        self._request("PUT", url, f"PUT {path}", payload=data)

    def _put_multipart(self, path: str, data: bytes):
        """CreateMultipartUpload -> UploadPart xN -> CompleteMultipartUpload
        (ref: object_io.rs multipart surface)."""
        url, _b, _k = self._url(path, "uploads")
        r = self._request("POST", url, f"mpu-create {path}")
        m = re.search(r"<UploadId>([^<]+)</UploadId>", r.text)
        if m is None:
            raise ObjectStoreError(f"multipart create: no UploadId: {r.text[:200]}")
        upload_id = m.group(1)
        etags = []
        part = 1
        for off in range(0, len(data), self.MULTIPART_CHUNK):
            chunk = data[off:off + self.MULTIPART_CHUNK]
            purl, _b2, _k2 = self._url(
                path, f"partNumber={part}&uploadId={upload_id}")
            pr = self._request("PUT", purl, f"mpu-part {path}#{part}",
                               payload=chunk)
            etags.append((part, pr.headers.get("ETag", f'"{part}"')))
            part += 1
        body = "<CompleteMultipartUpload>" + "".join(
            f"<Part><PartNumber>{p}</PartNumber><ETag>{t}</ETag></Part>"
            for p, t in etags) + "</CompleteMultipartUpload>"
        curl, _b3, _k3 = self._url(path, f"uploadId={upload_id}")
        self._request("POST", curl, f"mpu-complete {path}",
                      payload=body.encode())

    def list_prefix(self, path_prefix):
        parsed = urllib.parse.urlsplit(path_prefix)
        bucket = parsed.netloc
        prefix = parsed.path.lstrip("/")
        out = []
        token = None
        while True:
            q = ("list-type=2&prefix=" +
                 urllib.parse.quote(prefix, safe=""))
            if token:
                q += "&continuation-token=" + urllib.parse.quote(token,
                                                                 safe="")
            if self.cfg.endpoint_url:
                base = self.cfg.endpoint_url.rstrip("/")
                url = f"{base}/{bucket}?{q}"
            else:
                url = (f"https://{bucket}.s3.{self.cfg.region_name}"
                       f".amazonaws.com/?{q}")
            r = self._request("GET", url, f"LIST {path_prefix}")
            text = r.text
            for m in re.finditer(
                    r"<Contents>.*?<Key>([^<]+)</Key>.*?"
                    r"<Size>(\d+)</Size>.*?</Contents>", text, re.S):
                out.append((f"s3://{bucket}/{m.group(1)}",
                            int(m.group(2))))
            mt = re.search(r"<NextContinuationToken>([^<]+)</"
                           r"NextContinuationToken>", text)
            if mt is None:
                break
            token = mt.group(1)
        return out


# ---------------------------------------------------------------------------
# dispatch
# ---------------------------------------------------------------------------

_DEFAULT_IO_CONFIG: Optional[IOConfig] = None


def set_default_io_config(cfg: Optional[IOConfig]) -> None:
    global _DEFAULT_IO_CONFIG
    _DEFAULT_IO_CONFIG = cfg


def get_source(path: str,
               io_config: Optional[IOConfig] = None) -> ObjectSource:
    cfg = io_config or _DEFAULT_IO_CONFIG or IOConfig()
    scheme = urllib.parse.urlsplit(path).scheme
    if scheme in ("s3", "s3a"):
        return S3Source(cfg.s3)
    if scheme in ("http", "https"):
        return HTTPSource(cfg.http)
    if scheme in ("", "file"):
        return LocalSource()
    raise ValueError(f"unsupported object-store scheme {scheme!r} in {path}")


def is_remote(path: str) -> bool:
    return urllib.parse.urlsplit(path).scheme in ("s3", "s3a", "http",
                                                  "https")


def glob_paths(pattern: str,
               io_config: Optional[IOConfig] = None) -> List[str]:
    if is_remote(pattern):
        return get_source(pattern, io_config).glob(pattern)
    import glob as _g
    if any(ch in pattern for ch in "*?["):
        return sorted(_g.glob(pattern, recursive=True))
    return [pattern]
