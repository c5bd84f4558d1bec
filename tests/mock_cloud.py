"""In-process mock GCS (JSON API) and Azure Blob (REST) servers for the
cloud-backend tests, in the style of mock_s3.py (ref test pattern: the
reference's fake-gcs-server / Azurite integration fixtures,
tests/integration/io/conftest.py)."""
from __future__ import annotations

import json
import re
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class _MockBase:
    def __init__(self, handler_cls):
        self.objects: dict = {}          # "bucket/key" -> bytes
        self.fail_next = 0
        self.requests = 0
        self._server = ThreadingHTTPServer(("127.0.0.1", 0), handler_cls)
        self._server.mock = self  # type: ignore
        self.port = self._server.server_port
        self.endpoint = f"http://127.0.0.1:{self.port}"
        self._thread = threading.Thread(target=self._server.serve_forever,
                                        daemon=True)
        self._thread.start()

    def close(self):
        self._server.shutdown()


class MockGCS(_MockBase):
    """fake-gcs-server-style JSON API: objects get/list/upload incl. the
    resumable protocol; 2-item list pages to exercise pageToken paging."""

    PAGE = 2

    def __init__(self):
        self.sessions: dict = {}      # session id -> {"key":, "data": bytearray}
        self._sid = 0
        srv = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _fault(self) -> bool:
                srv.requests += 1
                if srv.fail_next > 0:
                    srv.fail_next -= 1
                    self.send_response(503)
                    self.end_headers()
                    return True
                return False

            def _send(self, code, body=b"", headers=()):
                self.send_response(code)
                for k, v in headers:
                    self.send_header(k, v)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                self.wfile.write(body)

            def do_GET(self):
                if self._fault():
                    return
                parsed = urllib.parse.urlsplit(self.path)
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                m = re.match(r"/storage/v1/b/([^/]+)/o/(.+)$", parsed.path)
                if m:
                    key = m.group(1) + "/" + urllib.parse.unquote(m.group(2))
                    if key not in srv.objects:
                        return self._send(404)
                    if qs.get("alt") == "media":
                        data = srv.objects[key]
                        rng = self.headers.get("Range")
                        code = 200
                        if rng:
                            mm = re.match(r"bytes=(\d+)-(\d+)", rng)
                            data = data[int(mm.group(1)):int(mm.group(2)) + 1]
                            code = 206
                        return self._send(code, data)
                    meta = {"name": key.split("/", 1)[1],
                            "size": str(len(srv.objects[key]))}
                    return self._send(200, json.dumps(meta).encode())
                m = re.match(r"/storage/v1/b/([^/]+)/o$", parsed.path)
                if m:
                    bucket = m.group(1)
                    prefix = qs.get("prefix", "")
                    items = sorted(
                        (k[len(bucket) + 1:], len(v))
                        for k, v in srv.objects.items()
                        if k.startswith(bucket + "/") and
                        k[len(bucket) + 1:].startswith(prefix))
                    start = int(qs.get("pageToken", 0) or 0)
                    page = items[start:start + srv.PAGE]
                    doc = {"items": [{"name": n, "size": str(s)}
                                     for n, s in page]}
                    if start + srv.PAGE < len(items):
                        doc["nextPageToken"] = str(start + srv.PAGE)
                    return self._send(200, json.dumps(doc).encode())
                self._send(400)

            def do_POST(self):
                if self._fault():
                    return
                parsed = urllib.parse.urlsplit(self.path)
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                n = int(self.headers.get("Content-Length", 0))
                body = self.rfile.read(n)
                m = re.match(r"/upload/storage/v1/b/([^/]+)/o$", parsed.path)
                if not m:
                    return self._send(400)
                key = m.group(1) + "/" + qs.get("name", "")
                if qs.get("uploadType") == "media":
                    srv.objects[key] = body
                    return self._send(200, b"{}")
                if qs.get("uploadType") == "resumable":
                    srv._sid += 1
                    sid = f"sess{srv._sid}"
                    srv.sessions[sid] = {"key": key, "data": bytearray()}
                    loc = f"http://127.0.0.1:{srv.port}/resumable/{sid}"
                    return self._send(200, b"{}", [("Location", loc)])
                self._send(400)

            def do_PUT(self):
                if self._fault():
                    return
                parsed = urllib.parse.urlsplit(self.path)
                m = re.match(r"/resumable/(.+)$", parsed.path)
                if not m or m.group(1) not in srv.sessions:
                    return self._send(404)
                sess = srv.sessions[m.group(1)]
                n = int(self.headers.get("Content-Length", 0))
                body = self.rfile.read(n)
                cr = self.headers.get("Content-Range", "")
                mm = re.match(r"bytes (\d+)-(\d+)/(\d+)", cr)
                sess["data"][int(mm.group(1)):int(mm.group(2)) + 1] = body
                if int(mm.group(2)) + 1 >= int(mm.group(3)):
                    srv.objects[sess["key"]] = bytes(sess["data"])
                    del srv.sessions[m.group(1)]
                    return self._send(200, b"{}")
                self._send(308)

        super().__init__(Handler)


class MockAzure(_MockBase):
    """Azurite-style Blob endpoint at /{account}/...; verifies the client's
    SharedKey Authorization header by recomputing the signature from the
    received request with the known account key."""

    ACCOUNT = "devacct"
    KEY_B64 = "c2VjcmV0LWtleS1mb3ItdGVzdHM="     # base64("secret-key-for-tests")
    PAGE = 2

    def __init__(self):
        self.blocks: dict = {}        # "cont/blob" -> {block_id: bytes}
        self.auth_failures = 0
        srv = self

        class Handler(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _fault(self) -> bool:
                srv.requests += 1
                if srv.fail_next > 0:
                    srv.fail_next -= 1
                    self.send_response(500)
                    self.end_headers()
                    return True
                return False

            def _send(self, code, body=b"", headers=()):
                self.send_response(code)
                for k, v in headers:
                    self.send_header(k, v)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                if self.command != "HEAD":
                    self.wfile.write(body)

            def _check_auth(self, method, body_len) -> bool:
                got = self.headers.get("Authorization")
                if not got:
                    return True          # anonymous allowed in tests
                from daft_amd.io.object_store import _azure_sharedkey_auth
                url = f"http://127.0.0.1:{srv.port}{self.path}"
                hdrs = {k: v for k, v in self.headers.items()
                        if k.lower().startswith("x-ms-") or
                        k in ("Content-Type", "Range")}
                want = _azure_sharedkey_auth(method, url, srv.ACCOUNT,
                                             srv.KEY_B64, hdrs, body_len)
                if got != want:
                    srv.auth_failures += 1
                    self._send(403, b"auth mismatch")
                    return False
                return True

            def _key(self, path):
                # /devacct/container/blob -> "container/blob"
                parts = path.lstrip("/").split("/", 1)
                return parts[1] if len(parts) > 1 else ""

            def do_PUT(self):
                if self._fault():
                    return
                parsed = urllib.parse.urlsplit(self.path)
                n = int(self.headers.get("Content-Length", 0))
                body = self.rfile.read(n)
                if not self._check_auth("PUT", n):
                    return
                key = self._key(parsed.path)
                key = urllib.parse.unquote(key)
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                if qs.get("comp") == "block":
                    srv.blocks.setdefault(key, {})[qs["blockid"]] = body
                    return self._send(201)
                if qs.get("comp") == "blocklist":
                    ids = re.findall(r"<Latest>([^<]+)</Latest>",
                                     body.decode())
                    blocks = srv.blocks.pop(key, {})
                    srv.objects[key] = b"".join(blocks[b] for b in ids)
                    return self._send(201)
                srv.objects[key] = body
                self._send(201)

            def do_GET(self):
                if self._fault():
                    return
                if not self._check_auth("GET", 0):
                    return
                parsed = urllib.parse.urlsplit(self.path)
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                if qs.get("comp") == "list":
                    # path is /devacct/container
                    container = parsed.path.lstrip("/").split("/")[1]
                    prefix = qs.get("prefix", "")
                    items = sorted(
                        (k[len(container) + 1:], len(v))
                        for k, v in srv.objects.items()
                        if k.startswith(container + "/") and
                        k[len(container) + 1:].startswith(prefix))
                    start = int(qs.get("marker", 0) or 0)
                    page = items[start:start + srv.PAGE]
                    xml = ["<EnumerationResults><Blobs>"]
                    for nme, sz in page:
                        xml.append(f"<Blob><Name>{nme}</Name><Properties>"
                                   f"<Content-Length>{sz}</Content-Length>"
                                   f"</Properties></Blob>")
                    xml.append("</Blobs>")
                    if start + srv.PAGE < len(items):
                        xml.append(f"<NextMarker>{start + srv.PAGE}"
                                   f"</NextMarker>")
                    xml.append("</EnumerationResults>")
                    return self._send(200, "".join(xml).encode())
                key = urllib.parse.unquote(self._key(parsed.path))
                if key not in srv.objects:
                    return self._send(404)
                data = srv.objects[key]
                rng = self.headers.get("Range")
                code = 200
                if rng:
                    mm = re.match(r"bytes=(\d+)-(\d+)", rng)
                    data = data[int(mm.group(1)):int(mm.group(2)) + 1]
                    code = 206
                self._send(code, data)

            def do_HEAD(self):
                if self._fault():
                    return
                if not self._check_auth("HEAD", 0):
                    return
                key = urllib.parse.unquote(
                    self._key(urllib.parse.urlsplit(self.path).path))
                if key not in srv.objects:
                    return self._send(404)
                self._send(200, srv.objects[key])

        super().__init__(Handler)
