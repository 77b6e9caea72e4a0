"""In-process fake WebHDFS namenode+datanode for connector tests.

Implements the protocol shape the connector depends on: LISTSTATUS /
GETFILESTATUS / MKDIRS / DELETE / RENAME answered directly, and the
OPEN / CREATE / APPEND namenode->datanode 307 redirect dance (the
"datanode" is the same server under /dn/)."""
from __future__ import annotations

import json
import threading
import urllib.parse
from http.server import BaseHTTPRequestHandler, ThreadingHTTPServer


class FakeWebHdfs:
    def __init__(self):
        self.files: dict[str, bytes] = {}
        self.dirs: set[str] = {"/"}
        self.requests: list[str] = []
        srv = self

        class H(BaseHTTPRequestHandler):
            def log_message(self, *a):
                pass

            def _path_op(self):
                u = urllib.parse.urlparse(self.path)
                q = dict(urllib.parse.parse_qsl(u.query))
                p = urllib.parse.unquote(u.path)
                dn = p.startswith("/dn")
                if dn:
                    p = p[len("/dn"):]
                assert p.startswith("/webhdfs/v1"), p
                p = p[len("/webhdfs/v1"):] or "/"
                p = p.rstrip("/") or "/"
                return p, q, dn

            def _json(self, obj, code=200):
                body = json.dumps(obj).encode()
                self.send_response(code)
                self.send_header("Content-Length", str(len(body)))
                self.send_header("Content-Type", "application/json")
                self.end_headers()
                self.wfile.write(body)

            def _redirect(self, code=307):
                loc = f"http://{srv.addr}/dn{self.path}"
                self.send_response(code)
                self.send_header("Location", loc)
                self.send_header("Content-Length", "0")
                self.end_headers()

            def _status_of(self, p):
                if p in srv.files:
                    return {"pathSuffix": "", "type": "FILE",
                            "length": len(srv.files[p]),
                            "modificationTime": 1700000000000}
                if p in srv.dirs:
                    return {"pathSuffix": "", "type": "DIRECTORY",
                            "length": 0, "modificationTime": 1700000000000}
                return None

            def do_GET(self):
                p, q, dn = self._path_op()
                op = q.get("op", "").upper()
                srv.requests.append(f"GET {op} {p}")
                if op == "LISTSTATUS":
                    if p not in srv.dirs:
                        st = self._status_of(p)
                        if st is None:
                            return self._json(
                                {"RemoteException": {
                                    "exception": "FileNotFoundException"}},
                                404)
                        return self._json(
                            {"FileStatuses": {"FileStatus": [st]}})
                    entries = []
                    prefix = p.rstrip("/") + "/"
                    seen = set()
                    for f in srv.files:
                        if f.startswith(prefix):
                            rest = f[len(prefix):]
                            name = rest.split("/")[0]
                            if "/" not in rest and name not in seen:
                                seen.add(name)
                                entries.append({
                                    "pathSuffix": name, "type": "FILE",
                                    "length": len(srv.files[f]),
                                    "modificationTime": 1700000000000})
                    for d in srv.dirs:
                        if d.startswith(prefix) and \
                                "/" not in d[len(prefix):] and d != p:
                            entries.append({
                                "pathSuffix": d[len(prefix):],
                                "type": "DIRECTORY", "length": 0,
                                "modificationTime": 1700000000000})
                    return self._json(
                        {"FileStatuses": {"FileStatus": entries}})
                if op == "GETFILESTATUS":
                    st = self._status_of(p)
                    if st is None:
                        return self._json(
                            {"RemoteException": {
                                "exception": "FileNotFoundException"}}, 404)
                    return self._json({"FileStatus": st})
                if op == "OPEN":
                    if not dn:
                        return self._redirect()
                    data = srv.files.get(p)
                    if data is None:
                        return self._json({}, 404)
                    off = int(q.get("offset", 0))
                    body = data[off:]
                    self.send_response(200)
                    self.send_header("Content-Length", str(len(body)))
                    self.end_headers()
                    self.wfile.write(body)
                    return
                self._json({}, 400)

            def _read_body(self):
                n = int(self.headers.get("Content-Length") or 0)
                return self.rfile.read(n) if n else b""

            def do_PUT(self):
                p, q, dn = self._path_op()
                op = q.get("op", "").upper()
                srv.requests.append(f"PUT {op} {p} dn={dn}")
                if op == "CREATE":
                    if not dn:
                        return self._redirect()
                    srv.files[p] = self._read_body()
                    parent = p.rsplit("/", 1)[0] or "/"
                    srv.dirs.add(parent)
                    return self._json({}, 201)
                if op == "MKDIRS":
                    srv.dirs.add(p.rstrip("/") or "/")
                    return self._json({"boolean": True})
                if op == "RENAME":
                    dst = q["destination"]
                    if p in srv.files:
                        srv.files[dst] = srv.files.pop(p)
                    return self._json({"boolean": True})
                self._json({}, 400)

            def do_POST(self):
                p, q, dn = self._path_op()
                op = q.get("op", "").upper()
                srv.requests.append(f"POST {op} {p} dn={dn}")
                if op == "APPEND":
                    if not dn:
                        return self._redirect()
                    srv.files[p] = srv.files.get(p, b"") + self._read_body()
                    return self._json({})
                self._json({}, 400)

            def do_DELETE(self):
                p, q, _ = self._path_op()
                srv.requests.append(f"DELETE {p}")
                if p in srv.files:
                    del srv.files[p]
                    return self._json({"boolean": True})
                if p in srv.dirs:
                    srv.dirs.discard(p)
                    if q.get("recursive") == "true":
                        pre = p.rstrip("/") + "/"
                        for f in [f for f in srv.files if f.startswith(pre)]:
                            del srv.files[f]
                    return self._json({"boolean": True})
                self._json({}, 404)

        self.httpd = ThreadingHTTPServer(("127.0.0.1", 0), H)
        self.addr = f"127.0.0.1:{self.httpd.server_port}"
        self._t = threading.Thread(target=self.httpd.serve_forever,
                                   daemon=True)
        self._t.start()

    def put(self, path: str, data: bytes) -> None:
        self.files[path] = data
        d = path.rsplit("/", 1)[0] or "/"
        while d != "/":
            self.dirs.add(d)
            d = d.rsplit("/", 1)[0] or "/"

    def stop(self):
        self.httpd.shutdown()
        self.httpd.server_close()
