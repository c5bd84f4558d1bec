"""In-process S3-compatible mock server for object-store tests
(ref test pattern: the reference's minio fixture + mock AWS server,
tests/integration/io/conftest.py:28-131, tests/io/mock_aws_server.py)."""
from __future__ import annotations

import re
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class MockS3:
    def __init__(self):
        self.objects: dict = {}      # "bucket/key" -> bytes
        self.uploads: dict = {}      # upload_id -> {part#: bytes}
        self.fail_next = 0           # fault injection: 500 the next N reqs
        self.requests = 0
        self._uid = 0
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
                    self.wfile.write(b"injected fault")
                    return True
                return False

            def do_PUT(self):
                if self._fault():
                    return
                n = int(self.headers.get("Content-Length", 0))
                body = self.rfile.read(n)
                parsed = urllib.parse.urlsplit(self.path)
                key = parsed.path.lstrip("/")
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                if "uploadId" in qs:
                    up = srv.uploads.setdefault(qs["uploadId"], {})
                    up[int(qs["partNumber"])] = body
                    self.send_response(200)
                    self.send_header("ETag", f'"p{qs["partNumber"]}"')
                    self.end_headers()
                    return
                srv.objects[key] = body
                self.send_response(200)
                self.send_header("ETag", '"x"')
                self.end_headers()

            def do_POST(self):
                if self._fault():
                    return
                parsed = urllib.parse.urlsplit(self.path)
                key = parsed.path.lstrip("/")
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                n = int(self.headers.get("Content-Length", 0))
                body = self.rfile.read(n)
                if "uploads" in parsed.query.split("&")[0] or \
                        "uploads" in qs:
                    srv._uid += 1
                    uid = f"up{srv._uid}"
                    srv.uploads[uid] = {}
                    xml = (f"<InitiateMultipartUploadResult><UploadId>{uid}"
                           f"</UploadId></InitiateMultipartUploadResult>")
                    self.send_response(200)
                    self.end_headers()
                    self.wfile.write(xml.encode())
                    return
                if "uploadId" in qs:
                    parts = srv.uploads.pop(qs["uploadId"], {})
                    srv.objects[key] = b"".join(
                        parts[i] for i in sorted(parts))
                    self.send_response(200)
                    self.end_headers()
                    self.wfile.write(b"<CompleteMultipartUploadResult/>")
                    return
                self.send_response(400)
                self.end_headers()

            def do_GET(self):
                if self._fault():
                    return
                parsed = urllib.parse.urlsplit(self.path)
                key = parsed.path.lstrip("/")
                qs = dict(urllib.parse.parse_qsl(parsed.query))
                if qs.get("list-type") == "2":
                    bucket = key.rstrip("/")
                    prefix = qs.get("prefix", "")
                    items = sorted(
                        (k[len(bucket) + 1:], len(v))
                        for k, v in srv.objects.items()
                        if k.startswith(bucket + "/") and
                        k[len(bucket) + 1:].startswith(prefix))
                    xml = ["<ListBucketResult>"]
                    for k, sz in items:
                        xml.append(f"<Contents><Key>{k}</Key>"
                                   f"<Size>{sz}</Size></Contents>")
                    xml.append("</ListBucketResult>")
                    body = "".join(xml).encode()
                    self.send_response(200)
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                    return
                if key not in srv.objects:
                    self.send_response(404)
                    self.end_headers()
                    return
                data = srv.objects[key]
                rng = self.headers.get("Range")
                status = 200
                if rng:
                    m = re.match(r"bytes=(\d+)-(\d+)", rng)
                    data = data[int(m.group(1)):int(m.group(2)) + 1]
                    status = 206
                self.send_response(status)
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()
                self.wfile.write(data)

            def do_HEAD(self):
                if self._fault():
                    return
                key = urllib.parse.urlsplit(self.path).path.lstrip("/")
                if key not in self.server.mock.objects:  # type: ignore
                    self.send_response(404)
                    self.end_headers()
                    return
                self.send_response(200)
                self.send_header("Content-Length",
                                 str(len(srv.objects[key])))
                self.end_headers()

        self._server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self._server.mock = self  # type: ignore
        self.port = self._server.server_port
        self.endpoint = f"http://127.0.0.1:{self.port}"
        self._thread = threading.Thread(target=self._server.serve_forever,
                                        daemon=True)
        self._thread.start()

    def close(self):
        self._server.shutdown()
