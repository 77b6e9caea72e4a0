"""HDFS UFS connector over the WebHDFS REST protocol.

The reference reaches hdfs:// through OpenDAL with an embedded JVM
(/root/reference/crates/adapters/curvine-ufs-opendal/src/lib.rs:271,735,
curvine-hdfs-jni/src/jni/jvm.rs).  This environment has no JVM, so the
MI355X build speaks WebHDFS/HttpFS directly (the namenode's REST surface,
enabled by default on modern HDFS): pure stdlib HTTP, including the
namenode -> datanode 307 redirect dance for OPEN/CREATE/APPEND.

URI forms: ``hdfs://host:port/base`` or ``webhdfs://host:port/base``
(port = the namenode HTTP port, 9870 by default).  Properties:
``user`` (user.name auth), ``timeout_ms``.
"""
from __future__ import annotations

import json
import urllib.error
import urllib.parse
import urllib.request
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.ufs.base import UfsReader, UfsWriter, UnderFs


class _NoRedirect(urllib.request.HTTPErrorProcessor):
    """Keep 307 responses (the datanode redirect carries the Location)."""

    def http_response(self, request, response):
        if response.code in (307, 201):
            return response
        return super().http_response(request, response)

    https_response = http_response


_opener = urllib.request.build_opener(_NoRedirect)


class _HdfsReader(UfsReader):
    def __init__(self, fs: "WebHdfsUfs", path: str, offset: int):
        self.fs = fs
        self.path = path
        self.pos = offset
        self._resp = None

    def _ensure(self):
        if self._resp is None:
            self._resp = self.fs._open_stream(self.path, self.pos)

    def read(self, size: int) -> bytes:
        self._ensure()
        data = self._resp.read(size)
        self.pos += len(data)
        return data

    def seek(self, offset: int) -> None:
        if offset != self.pos:
            self.close()
            self.pos = offset

    def close(self) -> None:
        if self._resp is not None:
            try:
                self._resp.close()
            except Exception:  # noqa: BLE001
                pass
            self._resp = None


class _HdfsWriter(UfsWriter):
    """CREATE (redirect) with the first buffer, APPEND for the rest —
    bounded memory for arbitrarily large objects."""

    CHUNK = 8 << 20

    def __init__(self, fs: "WebHdfsUfs", path: str):
        self.fs = fs
        self.path = path
        self.buf = bytearray()
        self.created = False

    def write(self, data: bytes) -> int:
        self.buf += data
        while len(self.buf) >= self.CHUNK:
            self._flush_chunk(bytes(self.buf[:self.CHUNK]))
            del self.buf[:self.CHUNK]
        return len(data)

    def _flush_chunk(self, chunk: bytes) -> None:
        if not self.created:
            self.fs._create(self.path, chunk)
            self.created = True
        else:
            self.fs._append(self.path, chunk)

    def close(self) -> None:
        if self.buf or not self.created:
            self._flush_chunk(bytes(self.buf))
            self.buf.clear()


class WebHdfsUfs(UnderFs):
    scheme = "hdfs"

    def __init__(self, uri: str, properties: dict | None = None):
        p = urllib.parse.urlparse(uri)
        if not p.netloc:
            raise err.InvalidArgument(f"hdfs uri needs host:port: {uri!r}")
        self.host = p.netloc
        self.base = p.path.rstrip("/")
        props = properties or {}
        self.user = props.get("user", "")
        self.timeout = int(props.get("timeout_ms", 30_000)) / 1000.0

    # ---------------- http plumbing ----------------
    def _url(self, path: str, op: str, host: str | None = None,
             **params) -> str:
        full = urllib.parse.quote(self.base + "/" + path.lstrip("/"))
        q = {"op": op, **params}
        if self.user:
            q["user.name"] = self.user
        return (f"http://{host or self.host}/webhdfs/v1{full}?"
                + urllib.parse.urlencode(q))

    def _req(self, method: str, url: str, data: bytes | None = None,
             stream: bool = False):
        req = urllib.request.Request(url, data=data, method=method)
        try:
            resp = _opener.open(req, timeout=self.timeout)
        except urllib.error.HTTPError as e:
            body = e.read().decode(errors="replace")
            if e.code == 404:
                raise err.FileNotFound(url.split("?")[0]) from e
            raise err.FsError(f"webhdfs {method} {e.code}: {body[:200]}") \
                from e
        except OSError as e:
            raise err.ConnectError(f"webhdfs {self.host}: {e}") from e
        if stream:
            return resp
        with resp:
            body = resp.read()
        return json.loads(body) if body else {}

    def _redirected(self, method: str, op: str, path: str, payload: bytes,
                    **params) -> None:
        """Two-step namenode -> datanode write (the WebHDFS protocol)."""
        url = self._url(path, op, **params)
        req = urllib.request.Request(url, method=method)
        try:
            resp = _opener.open(req, timeout=self.timeout)
        except urllib.error.HTTPError as e:
            raise err.FsError(f"webhdfs {op} {e.code}") from e
        with resp:
            loc = resp.headers.get("Location")
        if resp.code == 307 and loc:
            self._req(method, loc, data=payload)
        elif resp.code not in (200, 201):
            raise err.FsError(f"webhdfs {op} unexpected {resp.code}")

    # ---------------- UnderFs surface ----------------
    def list_files(self, path: str = "/", recursive: bool = True) -> list[dict]:
        out: list[dict] = []
        stack = [path.rstrip("/") or "/"]
        while stack:
            cur = stack.pop()
            r = self._req("GET", self._url(cur, "LISTSTATUS"))
            for st in r.get("FileStatuses", {}).get("FileStatus", []):
                name = st.get("pathSuffix", "")
                child = (cur.rstrip("/") + "/" + name) if name else cur
                if st.get("type") == "DIRECTORY":
                    if recursive:
                        stack.append(child)
                else:
                    out.append({"path": child, "length": st.get("length", 0),
                                "is_dir": False,
                                "mtime_ms": st.get("modificationTime", 0)})
        return out

    def status(self, path: str) -> Optional[dict]:
        try:
            r = self._req("GET", self._url(path, "GETFILESTATUS"))
        except err.FileNotFound:
            return None
        st = r.get("FileStatus", {})
        return {"path": path, "length": st.get("length", 0),
                "is_dir": st.get("type") == "DIRECTORY",
                "mtime_ms": st.get("modificationTime", 0)}

    def _open_stream(self, path: str, offset: int):
        params = {"offset": offset} if offset else {}
        url = self._url(path, "OPEN", **params)
        resp = self._req("GET", url, stream=True)
        if resp.code == 307:
            loc = resp.headers.get("Location")
            resp.close()
            return self._req("GET", loc, stream=True)
        return resp

    def open(self, path: str, offset: int = 0) -> UfsReader:
        return _HdfsReader(self, path, offset)

    def create(self, path: str) -> UfsWriter:
        return _HdfsWriter(self, path)

    def _create(self, path: str, data: bytes) -> None:
        self._redirected("PUT", "CREATE", path, data, overwrite="true")

    def _append(self, path: str, data: bytes) -> None:
        self._redirected("POST", "APPEND", path, data)

    def delete(self, path: str, recursive: bool = False) -> None:
        self._req("DELETE", self._url(path, "DELETE",
                                      recursive=str(recursive).lower()))

    def mkdir(self, path: str) -> None:
        self._req("PUT", self._url(path, "MKDIRS"))

    def rename(self, src: str, dst: str) -> None:
        dst_full = self.base + "/" + dst.lstrip("/")
        self._req("PUT", self._url(src, "RENAME", destination=dst_full))
