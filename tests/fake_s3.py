"""Minimal in-process S3 server for connector tests: object PUT/GET
(with Range)/HEAD/DELETE, ListObjectsV2 with pagination, and multipart
uploads.  Single bucket namespace; SigV4 presence is asserted, not
verified."""
from __future__ import annotations

import threading
import urllib.parse
import uuid
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class FakeS3:
    def __init__(self, max_keys: int = 3):
        self.objects: dict[str, bytes] = {}     # "bucket/key" -> bytes
        self.uploads: dict[str, dict[int, bytes]] = {}
        self.max_keys = max_keys
        # fault injection: fail the next N requests with this status
        # (throttling/5xx retry tests)
        self.fail_next = 0
        self.fail_status = 503
        self.requests_seen = 0
        store = self

        class Handler(BaseHTTPRequestHandler):
            protocol_version = "HTTP/1.1"

            def log_message(self, *a):
                pass

            def _key(self):
                parts = urllib.parse.urlsplit(self.path)
                return urllib.parse.unquote(parts.path.lstrip("/")), \
                    urllib.parse.parse_qs(parts.query,
                                          keep_blank_values=True)

            def _maybe_fail(self) -> bool:
                store.requests_seen += 1
                if store.fail_next > 0:
                    store.fail_next -= 1
                    self._reply(store.fail_status, b"injected failure")
                    return True
                return False

            def _reply(self, code: int, body: bytes = b"", headers=None):
                self.send_response(code)
                for k, v in (headers or {}).items():
                    self.send_header(k, v)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                if body:
                    self.wfile.write(body)

            def _body(self) -> bytes:
                n = int(self.headers.get("Content-Length", 0))
                return self.rfile.read(n) if n else b""

            def do_PUT(self):
                if self._maybe_fail():
                    return
                assert self.headers.get("authorization", "").startswith(
                    "AWS4-HMAC-SHA256"), "unsigned request"
                key, q = self._key()
                data = self._body()
                if "uploadId" in q:
                    up = store.uploads.get(q["uploadId"][0])
                    if up is None:
                        return self._reply(404)
                    pn = int(q["partNumber"][0])
                    up[pn] = data
                    return self._reply(200, headers={
                        "ETag": f'"etag-{pn}"'})
                store.objects[key] = data
                self._reply(200, headers={"ETag": '"etag"'})

            def do_POST(self):
                key, q = self._key()
                if "uploads" in q:
                    uid = uuid.uuid4().hex
                    store.uploads[uid] = {}
                    body = (f"<InitiateMultipartUploadResult>"
                            f"<UploadId>{uid}</UploadId>"
                            f"</InitiateMultipartUploadResult>").encode()
                    return self._reply(200, body)
                if "uploadId" in q:
                    self._body()
                    up = store.uploads.pop(q["uploadId"][0], None)
                    if up is None:
                        return self._reply(404)
                    store.objects[key] = b"".join(
                        up[i] for i in sorted(up))
                    return self._reply(
                        200, b"<CompleteMultipartUploadResult/>")
                self._reply(400)

            def do_GET(self):
                if self._maybe_fail():
                    return
                key, q = self._key()
                if "list-type" in q:
                    return self._list(key.rstrip("/"), q)
                data = store.objects.get(key)
                if data is None:
                    return self._reply(404)
                rng = self.headers.get("Range")
                if rng:
                    a, b = rng.split("=")[1].split("-")
                    a, b = int(a), min(int(b), len(data) - 1)
                    part = data[a:b + 1]
                    return self._reply(206, part, headers={
                        "Content-Range": f"bytes {a}-{b}/{len(data)}"})
                self._reply(200, data)

            def _list(self, bucket, q):
                prefix = q.get("prefix", [""])[0]
                token = q.get("continuation-token", [None])[0]
                keys = sorted(k for k in store.objects
                              if k.startswith(f"{bucket}/{prefix}"))
                if token:
                    keys = [k for k in keys if k > token]
                page, rest = keys[:store.max_keys], keys[store.max_keys:]
                items = "".join(
                    f"<Contents><Key>{k.split('/', 1)[1]}</Key>"
                    f"<Size>{len(store.objects[k])}</Size></Contents>"
                    for k in page)
                nxt = (f"<NextContinuationToken>{page[-1]}"
                       f"</NextContinuationToken>") if rest else ""
                body = (f"<ListBucketResult>{items}{nxt}"
                        f"</ListBucketResult>").encode()
                self._reply(200, body)

            def do_HEAD(self):
                # HEAD: Content-Length reflects the object, body omitted
                key, _ = self._key()
                data = store.objects.get(key)
                self.send_response(404 if data is None else 200)
                self.send_header("Content-Length",
                                 "0" if data is None else str(len(data)))
                self.end_headers()

            def do_DELETE(self):
                key, q = self._key()
                if "uploadId" in q:
                    store.uploads.pop(q["uploadId"][0], None)
                    return self._reply(204)
                store.objects.pop(key, None)
                self._reply(204)

        self.server = ThreadingHTTPServer(("127.0.0.1", 0), Handler)
        self.port = self.server.server_address[1]
        self.endpoint = f"http://127.0.0.1:{self.port}"
        self._thread = threading.Thread(target=self.server.serve_forever,
                                        daemon=True)

    def start(self) -> "FakeS3":
        self._thread.start()
        return self

    def stop(self) -> None:
        self.server.shutdown()
        self.server.server_close()
