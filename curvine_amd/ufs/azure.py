"""Azure Blob Storage UFS connector (abfs:// / az://).

The reference reaches Azure through OpenDAL
(/root/reference/crates/adapters/curvine-ufs-opendal/src/lib.rs); this
is a direct REST implementation on the stdlib HTTP client: SharedKey
request signing (the storage-account HMAC scheme), List Blobs XML
paging, ranged GET, Put Blob / Put Block + Put Block List for chunked
uploads, Delete, and server-side Copy for rename.

URI forms: ``az://container/prefix`` or
``abfs://container@account.blob.core.windows.net/prefix``.
Properties: ``account``, ``account_key`` (base64), ``endpoint``
(override for emulators/fakes, e.g. "http://127.0.0.1:10000/account").
"""
from __future__ import annotations

import base64
import hashlib
import hmac
import urllib.error
import urllib.parse
import urllib.request
import xml.etree.ElementTree as ET
from email.utils import formatdate
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.ufs.base import UfsReader, UfsWriter, UnderFs

_VER = "2021-08-06"


class _AzReader(UfsReader):
    def __init__(self, fs: "AzureUfs", name: str, offset: int, length: int,
                 chunk: int = 8 << 20):
        self.fs, self.name = fs, name
        self.pos, self.length, self.chunk = offset, length, chunk
        self._buf = b""

    def read(self, size: int) -> bytes:
        if self.pos >= self.length:
            return b""
        if len(self._buf) < size:
            start = self.pos + len(self._buf)
            end = min(start + max(size, self.chunk), self.length) - 1
            if start <= end:
                self._buf += self.fs._get_range(self.name, start, end)
        out, self._buf = self._buf[:size], self._buf[size:]
        self.pos += len(out)
        return out

    def seek(self, offset: int) -> None:
        self.pos = offset
        self._buf = b""


class _AzWriter(UfsWriter):
    """Put Block (staged, base64 ids) + Put Block List commit — bounded
    memory for large objects; single Put Blob for small ones."""

    def __init__(self, fs: "AzureUfs", name: str, block_size: int = 8 << 20):
        self.fs, self.name = fs, name
        self.block_size = block_size
        self._buf = bytearray()
        self._ids: list[str] = []

    def write(self, data: bytes) -> int:
        self._buf += data
        while len(self._buf) >= self.block_size:
            self._stage(bytes(self._buf[:self.block_size]))
            del self._buf[:self.block_size]
        return len(data)

    def _stage(self, chunk: bytes) -> None:
        bid = base64.b64encode(f"blk{len(self._ids):08d}".encode()).decode()
        self.fs._put_block(self.name, bid, chunk)
        self._ids.append(bid)

    def close(self) -> None:
        if not self._ids:
            self.fs._put_blob(self.name, bytes(self._buf))
        else:
            if self._buf:
                self._stage(bytes(self._buf))
            self.fs._put_block_list(self.name, self._ids)
        self._buf.clear()


class AzureUfs(UnderFs):
    scheme = "az"

    def __init__(self, uri: str, properties: dict | None = None):
        p = properties or {}
        rest = uri.split("://", 1)[1]
        if "@" in rest.split("/", 1)[0]:
            cont_at, _, prefix = rest.partition("/")
            self.container, host = cont_at.split("@", 1)
            self.account = p.get("account", host.split(".")[0])
            default_ep = f"https://{host}"
        else:
            self.container, _, prefix = rest.partition("/")
            self.account = p.get("account", "")
            default_ep = f"https://{self.account}.blob.core.windows.net"
        self.prefix = prefix.strip("/")
        self.endpoint = p.get("endpoint", default_ep).rstrip("/")
        self.key = base64.b64decode(p.get("account_key", "") or b"")
        self.timeout = int(p.get("timeout_ms", 30_000)) / 1000.0

    # ---------------- signing + http ----------------
    def _name(self, path: str) -> str:
        rel = path.strip("/")
        if not self.prefix:
            return rel
        return f"{self.prefix}/{rel}" if rel else self.prefix

    def _sign(self, method: str, path: str, query: dict, headers: dict,
              content_len: int) -> None:
        """SharedKey: HMAC-SHA256 over the canonicalized request."""
        if not self.key:
            return
        cr = f"/{self.account}/{self.container}"
        if path:
            cr += f"/{path}"
        for k in sorted(query):
            cr += f"\n{k}:{query[k]}"
        ch = "".join(f"{k}:{headers[k]}\n" for k in sorted(headers)
                     if k.startswith("x-ms-"))
        sts = "\n".join([
            method, "", "",
            str(content_len) if content_len else "",
            "", headers.get("Content-Type", ""), "", "", "", "", "", "",
        ]) + "\n" + ch + cr
        sig = base64.b64encode(
            hmac.new(self.key, sts.encode(), hashlib.sha256).digest()
        ).decode()
        headers["Authorization"] = f"SharedKey {self.account}:{sig}"

    def _req(self, method: str, path: str, query: dict | None = None,
             data: bytes = b"", headers: dict | None = None):
        query = query or {}
        headers = dict(headers or {})
        headers["x-ms-version"] = _VER
        headers["x-ms-date"] = formatdate(usegmt=True)
        self._sign(method, path, query, headers, len(data))
        qp = urllib.parse.urlencode(query)
        url = (f"{self.endpoint}/{self.container}"
               + (f"/{urllib.parse.quote(path)}" if path else "")
               + (f"?{qp}" if qp else ""))
        req = urllib.request.Request(url, data=data or None, method=method,
                                     headers=headers)
        try:
            return urllib.request.urlopen(req, timeout=self.timeout)
        except urllib.error.HTTPError as e:
            if e.code == 404:
                raise err.FileNotFound(path) from e
            body = e.read().decode(errors="replace")[:200]
            raise err.FsError(f"azure {method} {e.code}: {body}") from e
        except OSError as e:
            raise err.ConnectError(f"azure {self.endpoint}: {e}") from e

    # ---------------- blob ops ----------------
    def _get_range(self, name: str, start: int, end: int) -> bytes:
        with self._req("GET", name,
                       headers={"x-ms-range": f"bytes={start}-{end}"}) as r:
            return r.read()

    def _put_blob(self, name: str, data: bytes) -> None:
        self._req("PUT", name, data=data,
                  headers={"x-ms-blob-type": "BlockBlob"}).close()

    def _put_block(self, name: str, bid: str, data: bytes) -> None:
        self._req("PUT", name, {"comp": "block", "blockid": bid},
                  data=data).close()

    def _put_block_list(self, name: str, ids: list[str]) -> None:
        body = ("<?xml version='1.0' encoding='utf-8'?><BlockList>"
                + "".join(f"<Latest>{i}</Latest>" for i in ids)
                + "</BlockList>").encode()
        self._req("PUT", name, {"comp": "blocklist"}, data=body).close()

    # ---------------- UnderFs surface ----------------
    def list_files(self, path: str = "/", recursive: bool = True) -> list[dict]:
        want = self._name(path)
        prefix = (want + "/") if want else ""
        out, marker = [], ""
        while True:
            q = {"restype": "container", "comp": "list", "prefix": prefix}
            if marker:
                q["marker"] = marker
            with self._req("GET", "", q) as r:
                root = ET.fromstring(r.read())
            for b in root.iter("Blob"):
                name = b.findtext("Name", "")
                size = int(b.findtext("./Properties/Content-Length", "0"))
                rel = name[len(self.prefix):].lstrip("/") if self.prefix \
                    else name
                out.append({"path": "/" + rel, "length": size,
                            "is_dir": False})
            marker = root.findtext("NextMarker", "") or ""
            if not marker:
                break
        if not recursive:
            depth = want.count("/") + (1 if want else 0)
            out = [f for f in out if f["path"].strip("/").count("/") <= depth]
        return out

    def status(self, path: str) -> Optional[dict]:
        try:
            with self._req("HEAD", self._name(path)) as r:
                return {"path": path,
                        "length": int(r.headers.get("Content-Length", 0)),
                        "is_dir": False}
        except err.FileNotFound:
            # a "directory" exists if anything lives under it
            kids = self.list_files(path, recursive=True)
            return {"path": path, "length": 0, "is_dir": True} if kids \
                else None

    def open(self, path: str, offset: int = 0) -> UfsReader:
        st = self.status(path)
        if st is None or st["is_dir"]:
            raise err.FileNotFound(path)
        return _AzReader(self, self._name(path), offset, st["length"])

    def create(self, path: str) -> UfsWriter:
        return _AzWriter(self, self._name(path))

    def delete(self, path: str, recursive: bool = False) -> None:
        if recursive:
            for f in self.list_files(path, recursive=True):
                self._req("DELETE", self._name(f["path"])).close()
            return
        self._req("DELETE", self._name(path)).close()

    def mkdir(self, path: str) -> None:
        pass   # flat namespace: directories are implicit

    def rename(self, src: str, dst: str) -> None:
        sn, dn = self._name(src), self._name(dst)
        self._req("PUT", dn, headers={
            "x-ms-copy-source":
                f"{self.endpoint}/{self.container}/"
                + urllib.parse.quote(sn)}).close()
        self._req("DELETE", sn).close()
