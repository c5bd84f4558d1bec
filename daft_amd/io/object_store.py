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
from typing import List, Optional, Tuple


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
class GCSConfig:
    """Google Cloud Storage over its JSON API (ref:
    daft-io/src/google_cloud.rs + common/io-config/src/gcs.rs).  Auth is
    a static bearer token (config or GCS_TOKEN / GOOGLE_CLOUD_TOKEN env)
    or anonymous — the service-account JWT exchange needs RSA signing,
    which this offline image has no library for."""
    project_id: Optional[str] = None
    token: Optional[str] = None
    anonymous: bool = False
    endpoint_url: Optional[str] = None   # fake-gcs-server / emulators
    num_tries: int = 5
    retry_initial_backoff_ms: int = 100
    connect_timeout_ms: int = 10_000
    read_timeout_ms: int = 30_000

    def resolved(self) -> "GCSConfig":
        c = GCSConfig(**self.__dict__)
        c.token = c.token or os.environ.get("GCS_TOKEN") or \
            os.environ.get("GOOGLE_CLOUD_TOKEN")
        c.endpoint_url = c.endpoint_url or \
            os.environ.get("STORAGE_EMULATOR_HOST")
        return c


@dataclass
class AzureConfig:
    """Azure Blob Storage over the Blob REST API (ref:
    daft-io/src/azure_blob.rs + common/io-config/src/azure.rs).  Auth is
    SharedKey (account access key), a SAS token, or anonymous."""
    storage_account: Optional[str] = None
    access_key: Optional[str] = None
    sas_token: Optional[str] = None
    anonymous: bool = False
    endpoint_url: Optional[str] = None   # Azurite: http://host:port/acct
    num_tries: int = 5
    retry_initial_backoff_ms: int = 100
    connect_timeout_ms: int = 10_000
    read_timeout_ms: int = 30_000

    def resolved(self) -> "AzureConfig":
        c = AzureConfig(**self.__dict__)
        c.storage_account = c.storage_account or \
            os.environ.get("AZURE_STORAGE_ACCOUNT")
        c.access_key = c.access_key or os.environ.get("AZURE_STORAGE_KEY")
        c.sas_token = c.sas_token or os.environ.get("AZURE_STORAGE_SAS_TOKEN")
        return c


@dataclass
class HuggingFaceConfig:
    """Hugging Face Hub access (ref: common/io-config HFConfig)."""
    token: Optional[str] = None
    endpoint: Optional[str] = None       # mirrors / mock hubs
    anonymous: bool = False

    def resolved(self) -> "HuggingFaceConfig":
        c = HuggingFaceConfig(**self.__dict__)
        c.token = c.token or os.environ.get("HF_TOKEN")
        c.endpoint = c.endpoint or os.environ.get("HF_ENDPOINT")
        return c


@dataclass
class S3Credentials:
    """Static S3 credentials bundle (ref: daft.io.S3Credentials)."""
    key_id: str
    access_key: str
    session_token: Optional[str] = None
    expiry: Optional[object] = None


# vendor object-store configs mirrored for API parity; their backends
# need vendor SDKs/services this offline image cannot reach, so
# get_source() raises for their schemes (honest gating, not stubs
# pretending to work)
@dataclass
class CosConfig:
    region: Optional[str] = None
    key_id: Optional[str] = None
    access_key: Optional[str] = None


@dataclass
class TosConfig:
    region: Optional[str] = None
    key_id: Optional[str] = None
    access_key: Optional[str] = None


@dataclass
class GooseFSConfig:
    endpoint: Optional[str] = None


@dataclass
class HdfsConfig:
    namenode: Optional[str] = None
    port: int = 8020


@dataclass
class UnityConfig:
    endpoint: Optional[str] = None
    token: Optional[str] = None


@dataclass
class GravitinoConfig:
    uri: Optional[str] = None
    metalake: Optional[str] = None


@dataclass
class IOConfig:
    s3: S3Config = field(default_factory=S3Config)
    http: HTTPConfig = field(default_factory=HTTPConfig)
    gcs: GCSConfig = field(default_factory=GCSConfig)
    azure: AzureConfig = field(default_factory=AzureConfig)
    hf: HuggingFaceConfig = field(default_factory=HuggingFaceConfig)


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


class GCSSource(ObjectSource):
    """GCS JSON API: objects.get (alt=media, ranged), metadata, list with
    pageToken paging, media/resumable upload.  Works against real GCS
    (with a bearer token) and fake-gcs-server-style emulators.
    Ref: daft-io/src/google_cloud.rs."""

    RESUMABLE_CHUNK = 16 * 1024 * 1024

    def __init__(self, config: Optional[GCSConfig] = None):
        self.cfg = (config or GCSConfig()).resolved()
        import requests
        self._sess = requests.Session()

    def _base(self) -> str:
        ep = self.cfg.endpoint_url or "https://storage.googleapis.com"
        return ep.rstrip("/")

    def _split(self, path: str) -> Tuple[str, str]:
        parsed = urllib.parse.urlsplit(path)
        return parsed.netloc, parsed.path.lstrip("/")

    def _headers(self) -> dict:
        if self.cfg.anonymous or not self.cfg.token:
            return {}
        return {"Authorization": f"Bearer {self.cfg.token}"}

    def _request(self, method: str, url: str, what: str, payload=None,
                 extra_headers: Optional[dict] = None):
        def go():
            headers = self._headers()
            if extra_headers:
                headers.update(extra_headers)
            r = self._sess.request(
                method, url, data=payload, headers=headers,
                timeout=(self.cfg.connect_timeout_ms / 1000,
                         self.cfg.read_timeout_ms / 1000))
            if r.status_code == 404:
                raise NotFoundError(url)
            if r.status_code >= 400:
                raise ObjectStoreError(
                    f"GCS {r.status_code} for {what}: {r.text[:200]}")
            return r
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms, what)

    def _obj_url(self, bucket: str, obj: str, suffix: str = "") -> str:
        return (f"{self._base()}/storage/v1/b/{bucket}/o/"
                f"{urllib.parse.quote(obj, safe='')}{suffix}")

    def get(self, path, range_=None):
        bucket, obj = self._split(path)
        extra = {}
        if range_ is not None:
            extra["Range"] = f"bytes={range_[0]}-{range_[1] - 1}"
        return self._request("GET", self._obj_url(bucket, obj, "?alt=media"),
                             f"GET {path}", extra_headers=extra).content

    def get_size(self, path):
        import json
        bucket, obj = self._split(path)
        r = self._request("GET", self._obj_url(bucket, obj), f"STAT {path}")
        return int(json.loads(r.text)["size"])

    def put(self, path, data: bytes):
        bucket, obj = self._split(path)
        base = (f"{self._base()}/upload/storage/v1/b/{bucket}/o"
                f"?name={urllib.parse.quote(obj, safe='')}")
        if len(data) <= self.RESUMABLE_CHUNK:
            self._request("POST", base + "&uploadType=media", f"PUT {path}",
                          payload=data)
            return
        # resumable: one session POST, then sequential Content-Range PUTs
        r = self._request("POST", base + "&uploadType=resumable",
                          f"resumable-start {path}")
        session = r.headers.get("Location") or r.headers.get("location")
        if not session:
            raise ObjectStoreError(f"resumable upload: no session URI "
                                   f"for {path}")
        total = len(data)
        for off in range(0, total, self.RESUMABLE_CHUNK):
            chunk = data[off:off + self.RESUMABLE_CHUNK]
            hi = off + len(chunk) - 1
            def go(chunk=chunk, off=off, hi=hi):
                rr = self._sess.put(
                    session, data=chunk,
                    headers={"Content-Range":
                             f"bytes {off}-{hi}/{total}"},
                    timeout=(self.cfg.connect_timeout_ms / 1000,
                             self.cfg.read_timeout_ms / 1000))
                # 308 = chunk accepted, more expected
                if rr.status_code not in (200, 201, 308):
                    raise ObjectStoreError(
                        f"GCS {rr.status_code} for resumable {path}")
                return rr
            _with_retry(go, self.cfg.num_tries,
                        self.cfg.retry_initial_backoff_ms,
                        f"resumable-part {path}@{off}")

    def list_prefix(self, path_prefix):
        import json
        bucket, prefix = self._split(path_prefix)
        out = []
        token = None
        while True:
            q = "prefix=" + urllib.parse.quote(prefix, safe="")
            if token:
                q += "&pageToken=" + urllib.parse.quote(token, safe="")
            url = f"{self._base()}/storage/v1/b/{bucket}/o?{q}"
            doc = json.loads(self._request("GET", url,
                                           f"LIST {path_prefix}").text)
            for item in doc.get("items", []):
                out.append((f"gs://{bucket}/{item['name']}",
                            int(item.get("size", 0))))
            token = doc.get("nextPageToken")
            if not token:
                break
        return out


def _azure_sharedkey_auth(method: str, url: str, account: str, key_b64: str,
                          headers: dict, content_length: int) -> str:
    """SharedKey signature over the 2015+ canonical string
    (ref: daft-io/src/azure_blob.rs credential handling)."""
    import base64
    parsed = urllib.parse.urlsplit(url)
    # canonicalized x-ms-* headers, lowercase-sorted
    xms = sorted((k.lower(), v) for k, v in headers.items()
                 if k.lower().startswith("x-ms-"))
    canon_headers = "".join(f"{k}:{v}\n" for k, v in xms)
    # canonicalized resource: /account/path + sorted query params
    # (Azurite-style endpoints already embed /account in the path)
    path = parsed.path or "/"
    if path == f"/{account}" or path.startswith(f"/{account}/"):
        canon_res = path
    else:
        canon_res = f"/{account}{path}"
    for k, v in sorted(urllib.parse.parse_qsl(parsed.query,
                                              keep_blank_values=True)):
        canon_res += f"\n{k.lower()}:{v}"
    cl = str(content_length) if content_length else ""
    sts = "\n".join([
        method,
        "",                      # Content-Encoding
        "",                      # Content-Language
        cl,                      # Content-Length ("" when 0)
        "",                      # Content-MD5
        headers.get("Content-Type", ""),
        "",                      # Date (x-ms-date used instead)
        "", "", "", "",          # If-Modified/Match/None-Match/Unmodified
        headers.get("Range", ""),
        canon_headers + canon_res])
    sig = hmac.new(base64.b64decode(key_b64), sts.encode("utf-8"),
                   hashlib.sha256).digest()
    return f"SharedKey {account}:{base64.b64encode(sig).decode()}"


class AzureBlobSource(ObjectSource):
    """Azure Blob REST API: Get/Put Blob, Put Block + Put Block List for
    large uploads, List Blobs with marker paging.  az:// and abfs[s]://
    URIs name container/blob; the account comes from AzureConfig.
    Ref: daft-io/src/azure_blob.rs."""

    BLOCK_CHUNK = 16 * 1024 * 1024
    API_VERSION = "2021-08-06"

    def __init__(self, config: Optional[AzureConfig] = None):
        self.cfg = (config or AzureConfig()).resolved()
        import requests
        self._sess = requests.Session()

    def _base(self) -> str:
        if self.cfg.endpoint_url:
            return self.cfg.endpoint_url.rstrip("/")
        acct = self.cfg.storage_account
        if not acct:
            raise ObjectStoreError(
                "AzureConfig.storage_account (or AZURE_STORAGE_ACCOUNT) "
                "is required for az:// paths")
        return f"https://{acct}.blob.core.windows.net"

    def _split(self, path: str) -> Tuple[str, str]:
        parsed = urllib.parse.urlsplit(path)
        container = parsed.netloc
        # abfss://container@account.dfs.core.windows.net/blob form
        if "@" in container:
            container, host = container.split("@", 1)
            if self.cfg.storage_account is None:
                self.cfg.storage_account = host.split(".", 1)[0]
        return container, parsed.path.lstrip("/")

    def _url(self, container: str, blob: str, query: str = "") -> str:
        url = f"{self._base()}/{container}"
        if blob:
            url += "/" + urllib.parse.quote(blob)
        if query:
            sep = "?"
            if self.cfg.sas_token:
                url += "?" + self.cfg.sas_token.lstrip("?")
                sep = "&"
            url += sep + query
        elif self.cfg.sas_token:
            url += "?" + self.cfg.sas_token.lstrip("?")
        return url

    def _request(self, method: str, url: str, what: str, payload: bytes = b"",
                 extra_headers: Optional[dict] = None):
        def go():
            now = _dt.datetime.now(_dt.timezone.utc)
            headers = {"x-ms-date": now.strftime("%a, %d %b %Y %H:%M:%S GMT"),
                       "x-ms-version": self.API_VERSION}
            if extra_headers:
                headers.update(extra_headers)
            if self.cfg.access_key and not self.cfg.anonymous and \
                    not self.cfg.sas_token:
                headers["Authorization"] = _azure_sharedkey_auth(
                    method, url, self.cfg.storage_account or "",
                    self.cfg.access_key, headers, len(payload))
            r = self._sess.request(
                method, url, data=payload if payload else None,
                headers=headers,
                timeout=(self.cfg.connect_timeout_ms / 1000,
                         self.cfg.read_timeout_ms / 1000))
            if r.status_code == 404:
                raise NotFoundError(url)
            if r.status_code >= 400:
                raise ObjectStoreError(
                    f"Azure {r.status_code} for {what}: {r.text[:200]}")
            return r
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms, what)

    def get(self, path, range_=None):
        container, blob = self._split(path)
        extra = {}
        if range_ is not None:
            extra["Range"] = f"bytes={range_[0]}-{range_[1] - 1}"
        return self._request("GET", self._url(container, blob),
                             f"GET {path}", extra_headers=extra).content

    def get_size(self, path):
        container, blob = self._split(path)
        r = self._request("HEAD", self._url(container, blob), f"HEAD {path}")
        return int(r.headers["Content-Length"])

    def put(self, path, data: bytes):
        import base64
        container, blob = self._split(path)
        if len(data) <= self.BLOCK_CHUNK:
            self._request("PUT", self._url(container, blob), f"PUT {path}",
                          payload=data,
                          extra_headers={"x-ms-blob-type": "BlockBlob"})
            return
        ids = []
        for i, off in enumerate(range(0, len(data), self.BLOCK_CHUNK)):
            bid = base64.b64encode(f"blk{i:08d}".encode()).decode()
            q = ("comp=block&blockid=" +
                 urllib.parse.quote(bid, safe=""))
            self._request("PUT", self._url(container, blob, q),
                          f"put-block {path}#{i}",
                          payload=data[off:off + self.BLOCK_CHUNK])
            ids.append(bid)
        body = ("<?xml version='1.0' encoding='utf-8'?><BlockList>" +
                "".join(f"<Latest>{b}</Latest>" for b in ids) +
                "</BlockList>").encode()
        self._request("PUT", self._url(container, blob, "comp=blocklist"),
                      f"put-blocklist {path}", payload=body,
                      extra_headers={"Content-Type": "application/xml"})

    def list_prefix(self, path_prefix):
        parsed = urllib.parse.urlsplit(path_prefix)
        container = parsed.netloc.split("@", 1)[0]
        prefix = parsed.path.lstrip("/")
        scheme = parsed.scheme or "az"
        out = []
        marker = None
        while True:
            q = ("restype=container&comp=list&prefix=" +
                 urllib.parse.quote(prefix, safe=""))
            if marker:
                q += "&marker=" + urllib.parse.quote(marker, safe="")
            r = self._request("GET", self._url(container, "", q),
                              f"LIST {path_prefix}")
            text = r.text
            for m in re.finditer(
                    r"<Blob>.*?<Name>([^<]+)</Name>.*?"
                    r"<Content-Length>(\d+)</Content-Length>.*?</Blob>",
                    text, re.S):
                out.append((f"{scheme}://{container}/{m.group(1)}",
                            int(m.group(2))))
            mt = re.search(r"<NextMarker>([^<]+)</NextMarker>", text)
            if mt is None:
                break
            marker = mt.group(1)
        return out


class HuggingFaceSource(ObjectSource):
    """hf://{datasets|models|spaces}/owner/repo[@revision]/path over the
    Hub's resolve + tree APIs (ref: daft-io/src/huggingface/{mod,path}.rs).
    Auth: Bearer token from HFConfig-style env HF_TOKEN; endpoint
    overridable via HF_ENDPOINT (mock servers / mirrors).  Read-only."""

    def __init__(self, config=None, http_config: Optional[HTTPConfig] = None):
        self.cfg = http_config or HTTPConfig()
        hf = (config or HuggingFaceConfig()).resolved() \
            if not isinstance(config, HTTPConfig) else \
            HuggingFaceConfig().resolved()
        if isinstance(config, HTTPConfig):
            self.cfg = config
        import requests
        self._sess = requests.Session()
        self.endpoint = (hf.endpoint or
                         "https://huggingface.co").rstrip("/")
        self.token = None if hf.anonymous else hf.token

    def _split(self, path: str):
        parsed = urllib.parse.urlsplit(path)
        parts = (parsed.netloc + parsed.path).split("/")
        if len(parts) < 3:
            raise ValueError(f"hf:// path needs repo_type/owner/repo: "
                             f"{path}")
        repo_type, owner, repo = parts[0], parts[1], parts[2]
        rev = "main"
        if "@" in repo:
            repo, rev = repo.split("@", 1)
        return repo_type, f"{owner}/{repo}", rev, "/".join(parts[3:])

    def _resolve_url(self, path: str) -> str:
        repo_type, repo, rev, file = self._split(path)
        prefix = "" if repo_type == "models" else f"{repo_type}/"
        return (f"{self.endpoint}/{prefix}{repo}/resolve/"
                f"{urllib.parse.quote(rev)}/{file}")

    def _headers(self) -> dict:
        return {"Authorization": f"Bearer {self.token}"} if self.token \
            else {}

    def get(self, path, range_=None):
        url = self._resolve_url(path)
        def go():
            headers = self._headers()
            if range_ is not None:
                headers["Range"] = f"bytes={range_[0]}-{range_[1] - 1}"
            r = self._sess.get(url, headers=headers,
                               timeout=self.cfg.read_timeout_ms / 1000)
            if r.status_code == 404:
                raise NotFoundError(path)
            if r.status_code >= 400:
                raise ObjectStoreError(f"HF {r.status_code} for {path}")
            return r.content
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms, f"GET {path}")

    def get_size(self, path):
        url = self._resolve_url(path)
        def go():
            r = self._sess.head(url, headers=self._headers(),
                                allow_redirects=True,
                                timeout=self.cfg.read_timeout_ms / 1000)
            if r.status_code == 404:
                raise NotFoundError(path)
            n = r.headers.get("X-Linked-Size") or \
                r.headers.get("Content-Length")
            if r.status_code >= 400 or n is None:
                raise ObjectStoreError(f"HF HEAD {r.status_code} {path}")
            return int(n)
        return _with_retry(go, self.cfg.num_tries,
                           self.cfg.retry_initial_backoff_ms,
                           f"HEAD {path}")

    def put(self, path, data):
        raise ObjectStoreError("hf:// is read-only here (upload via the "
                               "huggingface_hub client)")

    def list_prefix(self, path_prefix):
        import json
        repo_type, repo, rev, prefix = self._split(path_prefix)
        api_type = repo_type if repo_type != "models" else "models"
        base = (f"{self.endpoint}/api/{api_type}/{repo}/tree/"
                f"{urllib.parse.quote(rev)}")
        out = []

        def walk(sub: str):
            url = base + (f"/{sub}" if sub else "") + "?recursive=true"
            def go():
                r = self._sess.get(url, headers=self._headers(),
                                   timeout=self.cfg.read_timeout_ms / 1000)
                if r.status_code == 404:
                    raise NotFoundError(path_prefix)
                if r.status_code >= 400:
                    raise ObjectStoreError(f"HF tree {r.status_code}")
                return r
            r = _with_retry(go, self.cfg.num_tries,
                            self.cfg.retry_initial_backoff_ms,
                            f"LIST {path_prefix}")
            for item in json.loads(r.text):
                if item.get("type") == "file":
                    p = item["path"]
                    at = "" if rev == "main" else f"@{rev}"
                    out.append((f"hf://{repo_type}/{repo}{at}/{p}",
                                int(item.get("size", 0))))

        # list from the deepest directory of the static prefix, then
        # keep only entries under the full requested prefix
        walk(prefix.rsplit("/", 1)[0] if "/" in prefix else "")
        at = "" if rev == "main" else f"@{rev}"
        want = f"hf://{repo_type}/{repo}{at}/{prefix}"
        return [(p, sz) for (p, sz) in out if p.startswith(want)]


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
    if scheme in ("gs", "gcs"):
        return GCSSource(cfg.gcs)
    if scheme in ("az", "abfs", "abfss", "wasb", "wasbs"):
        return AzureBlobSource(cfg.azure)
    if scheme == "hf":
        return HuggingFaceSource(getattr(cfg, "hf", None),
                                 http_config=cfg.http)
    if scheme in ("http", "https"):
        return HTTPSource(cfg.http)
    if scheme in ("", "file"):
        return LocalSource()
    raise ValueError(f"unsupported object-store scheme {scheme!r} in {path}")


_REMOTE_SCHEMES = ("s3", "s3a", "gs", "gcs", "az", "abfs", "abfss",
                   "wasb", "wasbs", "hf", "http", "https")


def is_remote(path: str) -> bool:
    return urllib.parse.urlsplit(path).scheme in _REMOTE_SCHEMES


def glob_paths(pattern: str,
               io_config: Optional[IOConfig] = None) -> List[str]:
    if is_remote(pattern):
        return get_source(pattern, io_config).glob(pattern)
    import glob as _g
    if any(ch in pattern for ch in "*?["):
        return sorted(_g.glob(pattern, recursive=True))
    return [pattern]
