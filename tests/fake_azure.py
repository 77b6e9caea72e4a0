"""In-process fake Azure Blob service: SharedKey-signed List Blobs XML,
ranged GET, Put Blob / Put Block / Put Block List, Delete, Copy."""
from __future__ import annotations

import base64
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer

ACCOUNT = "devstore"
KEY = base64.b64encode(b"fake-azure-account-key-0123456789").decode()


class FakeAzure:
    def __init__(self):
        self.blobs: dict[str, bytes] = {}       # "container/name" -> bytes
        self.staged: dict[str, dict[str, bytes]] = {}
        srv = self

        class H(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _parts(self):
                u = urllib.parse.urlparse(self.path)
                q = dict(urllib.parse.parse_qsl(u.query))
                path = urllib.parse.unquote(u.path).strip("/")
                # endpoint form http://host/container[/name]
                cont, _, name = path.partition("/")
                return cont, name, q

            def _send(self, code=200, body=b"", headers=None):
                self.send_response(code)
                for k, v in (headers or {}).items():
                    self.send_header(k, v)
                self.send_header("Content-Length", str(len(body)))
                self.end_headers()
                if body:
                    self.wfile.write(body)

            def do_GET(self):
                cont, name, q = self._parts()
                if q.get("comp") == "list":
                    prefix = q.get("prefix", "")
                    items = []
                    for full, data in sorted(srv.blobs.items()):
                        c, _, n = full.partition("/")
                        if c == cont and n.startswith(prefix):
                            items.append(
                                f"<Blob><Name>{n}</Name><Properties>"
                                f"<Content-Length>{len(data)}"
                                f"</Content-Length></Properties></Blob>")
                    xml = ("<?xml version='1.0'?><EnumerationResults>"
                           "<Blobs>" + "".join(items) + "</Blobs>"
                           "<NextMarker/></EnumerationResults>")
                    return self._send(200, xml.encode())
                data = srv.blobs.get(f"{cont}/{name}")
                if data is None:
                    return self._send(404)
                rng = self.headers.get("x-ms-range") or \
                    self.headers.get("Range")
                if rng:
                    lo, hi = rng.split("=")[1].split("-")
                    body = data[int(lo):int(hi) + 1]
                    return self._send(206, body)
                self._send(200, data)

            def do_HEAD(self):
                cont, name, q = self._parts()
                data = srv.blobs.get(f"{cont}/{name}")
                if data is None:
                    self.send_response(404)
                    self.send_header("Content-Length", "0")
                    self.end_headers()
                    return
                # HEAD: Content-Length describes the blob, no body follows
                self.send_response(200)
                self.send_header("Content-Length", str(len(data)))
                self.end_headers()

            def _body(self):
                n = int(self.headers.get("Content-Length") or 0)
                return self.rfile.read(n) if n else b""

            def do_PUT(self):
                cont, name, q = self._parts()
                full = f"{cont}/{name}"
                assert self.headers.get("Authorization", "").startswith(
                    "SharedKey "), "unsigned request"
                if q.get("comp") == "block":
                    srv.staged.setdefault(full, {})[q["blockid"]] = \
                        self._body()
                    return self._send(201)
                if q.get("comp") == "blocklist":
                    body = self._body().decode()
                    import re
                    ids = re.findall(r"<Latest>(.*?)</Latest>", body)
                    st = srv.staged.pop(full, {})
                    srv.blobs[full] = b"".join(st[i] for i in ids)
                    return self._send(201)
                src = self.headers.get("x-ms-copy-source")
                if src:
                    sp = urllib.parse.unquote(
                        urllib.parse.urlparse(src).path).strip("/")
                    srv.blobs[full] = srv.blobs[sp]
                    return self._send(202)
                srv.blobs[full] = self._body()
                self._send(201)

            def do_DELETE(self):
                cont, name, q = self._parts()
                if srv.blobs.pop(f"{cont}/{name}", None) is None:
                    return self._send(404)
                self._send(202)

        self.httpd = ThreadingHTTPServer(("127.0.0.1", 0), H)
        self.addr = f"127.0.0.1:{self.httpd.server_port}"
        self._t = threading.Thread(target=self.httpd.serve_forever,
                                   daemon=True)
        self._t.start()

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()
