"""S3 UFS adapter over the REST API with SigV4 signing.

Analog of the reference's opendal S3 under-filesystem
(/root/reference/crates/adapters/curvine-ufs-opendal/src/lib.rs:271,735).
Implemented directly over HTTP (requests) — no SDK in this image.  The
SigV4 canonical-request construction is unit-tested offline; live S3 needs
an endpoint reachable from the deployment.
"""
from __future__ import annotations

import datetime
import hashlib
import time
import random
import hmac
import urllib.parse
import xml.etree.ElementTree as ET
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.ufs.base import UfsReader, UfsWriter, UnderFs

try:
    import requests
except ImportError:  # pragma: no cover
    requests = None

_EMPTY_SHA = hashlib.sha256(b"").hexdigest()


def _sign(key: bytes, msg: str) -> bytes:
    return hmac.new(key, msg.encode(), hashlib.sha256).digest()


def sigv4_headers(method: str, url: str, region: str, access_key: str,
                  secret_key: str, payload_sha: str = _EMPTY_SHA,
                  now: Optional[datetime.datetime] = None,
                  extra_headers: Optional[dict] = None) -> dict:
    """Build AWS SigV4 Authorization headers for one request."""
    parts = urllib.parse.urlsplit(url)
    host = parts.netloc
    now = now or datetime.datetime.now(datetime.timezone.utc)
    amz_date = now.strftime("%Y%m%dT%H%M%SZ")
    datestamp = now.strftime("%Y%m%d")
    headers = {"host": host, "x-amz-date": amz_date,
               "x-amz-content-sha256": payload_sha}
    if extra_headers:
        headers.update({k.lower(): v for k, v in extra_headers.items()})
    signed = ";".join(sorted(headers))
    canonical_headers = "".join(f"{k}:{headers[k]}\n" for k in sorted(headers))
    # canonical query: sorted, URL-encoded
    q = urllib.parse.parse_qsl(parts.query, keep_blank_values=True)
    cq = "&".join(f"{urllib.parse.quote(k, safe='-_.~')}={urllib.parse.quote(v, safe='-_.~')}"
                  for k, v in sorted(q))
    creq = "\n".join([method, urllib.parse.quote(parts.path or "/", safe="/-_.~"),
                      cq, canonical_headers, signed, payload_sha])
    scope = f"{datestamp}/{region}/s3/aws4_request"
    sts = "\n".join(["AWS4-HMAC-SHA256", amz_date, scope,
                     hashlib.sha256(creq.encode()).hexdigest()])
    k = _sign(_sign(_sign(_sign(b"AWS4" + secret_key.encode(), datestamp),
                          region), "s3"), "aws4_request")
    sig = hmac.new(k, sts.encode(), hashlib.sha256).hexdigest()
    headers["authorization"] = (
        f"AWS4-HMAC-SHA256 Credential={access_key}/{scope}, "
        f"SignedHeaders={signed}, Signature={sig}")
    return headers


class _S3Reader(UfsReader):
    def __init__(self, fs: "S3Ufs", key: str, offset: int, length: int,
                 chunk: int = 8 << 20):
        self.fs, self.key, self.pos, self.length, self.chunk = fs, key, offset, length, chunk
        self._buf = b""

    def read(self, size: int) -> bytes:
        if self.pos >= self.length:
            return b""
        if len(self._buf) < size:
            want = max(size, self.chunk)
            end = min(self.pos + len(self._buf) + want, self.length) - 1
            start = self.pos + len(self._buf)
            if start <= end:
                self._buf += self.fs._get_range(self.key, start, end)
        out, self._buf = self._buf[:size], self._buf[size:]
        self.pos += len(out)
        return out

    def seek(self, offset: int) -> None:
        self.pos = offset
        self._buf = b""


class _S3Writer(UfsWriter):
    """Objects below ``part_size`` go up as one PUT; larger ones switch
    to multipart (CreateMultipartUpload / UploadPart / Complete), so a
    cache write-back never buffers more than one part in memory."""

    def __init__(self, fs: "S3Ufs", key: str, part_size: int = 8 << 20):
        self.fs, self.key = fs, key
        self.part_size = part_size
        self._buf = bytearray()
        self.upload_id: Optional[str] = None
        self._etags: list[str] = []

    def write(self, data: bytes) -> int:
        self._buf.extend(data)
        while len(self._buf) >= self.part_size:
            self._flush_part(self.part_size)
        return len(data)

    def _flush_part(self, n: int) -> None:
        if self.upload_id is None:
            self.upload_id = self.fs._create_multipart(self.key)
        chunk = bytes(self._buf[:n])
        del self._buf[:n]
        etag = self.fs._upload_part(self.key, self.upload_id,
                                    len(self._etags) + 1, chunk)
        self._etags.append(etag)

    def close(self) -> None:
        if self.upload_id is None:
            self.fs._put(self.key, bytes(self._buf))
            return
        if self._buf:
            self._flush_part(len(self._buf))
        self.fs._complete_multipart(self.key, self.upload_id, self._etags)

    def abort(self) -> None:
        if self.upload_id is not None:
            self.fs._abort_multipart(self.key, self.upload_id)


class S3Ufs(UnderFs):
    scheme = "s3"

    def __init__(self, uri: str, properties: dict):
        if requests is None:
            raise err.Unsupported("requests not available for S3 ufs")
        rest = uri.split("://", 1)[1]
        self.bucket, _, self.prefix = rest.partition("/")
        self.prefix = ("/" + self.prefix).rstrip("/")
        self.endpoint = properties.get("endpoint", f"https://s3.amazonaws.com")
        self.region = properties.get("region", "us-east-1")
        self.access_key = properties.get("access_key", "")
        self.secret_key = properties.get("secret_key", "")
        self.part_size = int(properties.get("multipart_part_size", 8 << 20))
        self.session = requests.Session()

    def _url(self, key: str, query: str = "") -> str:
        path = urllib.parse.quote(f"/{self.bucket}{key}")
        return f"{self.endpoint}{path}" + (f"?{query}" if query else "")

    def _key(self, path: str) -> str:
        return f"{self.prefix}/{path.strip('/')}"

    # transient statuses S3 documents for client retry (throttling and
    # internal errors), plus connection-level failures
    _RETRY_STATUS = frozenset({429, 500, 502, 503, 504})
    _RETRIES = 4

    def _req(self, method: str, url: str, data: bytes = b"",
             headers: Optional[dict] = None):
        payload_sha = hashlib.sha256(data).hexdigest()
        last: Exception = err.UfsError(f"s3 {method} {url}: no attempt")
        for attempt in range(self._RETRIES + 1):
            if attempt:
                # exponential backoff with jitter, capped at 8 s
                time.sleep(min(8.0, (2.0 ** (attempt - 1)) * 0.25)
                           * (0.5 + random.random()))
            hdrs = sigv4_headers(method, url, self.region, self.access_key,
                                 self.secret_key, payload_sha,
                                 extra_headers=headers)
            try:
                r = self.session.request(method, url, data=data or None,
                                         headers=hdrs, timeout=60)
            except Exception as e:  # noqa: BLE001 — conn reset/timeout
                last = err.UfsError(f"s3 {method} {url}: {e}")
                continue
            if r.status_code == 404:
                raise err.FileNotFound(url)
            if r.status_code in self._RETRY_STATUS:
                last = err.UfsError(f"s3 {method} {url}: {r.status_code} "
                                    f"{r.text[:200]}")
                continue
            if r.status_code >= 300:
                raise err.UfsError(f"s3 {method} {url}: {r.status_code} "
                                   f"{r.text[:200]}")
            return r
        raise last

    def _get_range(self, key: str, start: int, end: int) -> bytes:
        r = self._req("GET", self._url(key), headers={"range": f"bytes={start}-{end}"})
        return r.content

    def _put(self, key: str, data: bytes) -> None:
        self._req("PUT", self._url(key), data=data)

    # ---------------- multipart upload ----------------
    def _create_multipart(self, key: str) -> str:
        r = self._req("POST", self._url(key, "uploads="))
        root = ET.fromstring(r.content)
        ns = root.tag.split("}")[0] + "}" if "}" in root.tag else ""
        return root.find(f"{ns}UploadId").text

    def _upload_part(self, key: str, upload_id: str, part_no: int,
                     data: bytes) -> str:
        q = f"partNumber={part_no}&uploadId={urllib.parse.quote(upload_id)}"
        r = self._req("PUT", self._url(key, q), data=data)
        return r.headers.get("etag", "").strip('"')

    def _complete_multipart(self, key: str, upload_id: str,
                            etags: list[str]) -> None:
        body = "<CompleteMultipartUpload>" + "".join(
            f"<Part><PartNumber>{i + 1}</PartNumber><ETag>\"{e}\"</ETag></Part>"
            for i, e in enumerate(etags)) + "</CompleteMultipartUpload>"
        q = f"uploadId={urllib.parse.quote(upload_id)}"
        self._req("POST", self._url(key, q), data=body.encode())

    def _abort_multipart(self, key: str, upload_id: str) -> None:
        q = f"uploadId={urllib.parse.quote(upload_id)}"
        self._req("DELETE", self._url(key, q))

    def list_files(self, path: str = "/", recursive: bool = True) -> list[dict]:
        prefix = self._key(path).strip("/")
        if prefix:
            prefix += "/"
        out, token = [], None
        while True:
            q = f"list-type=2&prefix={urllib.parse.quote(prefix)}"
            if not recursive:
                q += "&delimiter=%2F"
            if token:
                q += f"&continuation-token={urllib.parse.quote(token)}"
            r = self._req("GET", f"{self.endpoint}/{self.bucket}?{q}")
            root = ET.fromstring(r.content)
            ns = root.tag.split("}")[0] + "}" if "}" in root.tag else ""
            for c in root.findall(f"{ns}Contents"):
                k = c.find(f"{ns}Key").text
                size = int(c.find(f"{ns}Size").text)
                rel = "/" + k[len(self.prefix.lstrip('/')):].lstrip("/")
                out.append({"path": rel, "length": size, "is_dir": False})
            token_el = root.find(f"{ns}NextContinuationToken")
            token = token_el.text if token_el is not None else None
            if not token:
                return out

    def status(self, path: str) -> Optional[dict]:
        try:
            r = self._req("HEAD", self._url(self._key(path)))
        except err.FileNotFound:
            return None
        return {"path": "/" + path.strip("/"),
                "length": int(r.headers.get("content-length", 0)),
                "is_dir": False}

    def open(self, path: str, offset: int = 0) -> UfsReader:
        st = self.status(path)
        if st is None:
            raise err.FileNotFound(path)
        return _S3Reader(self, self._key(path), offset, st["length"])

    def create(self, path: str) -> UfsWriter:
        return _S3Writer(self, self._key(path), self.part_size)

    def delete(self, path: str, recursive: bool = False) -> None:
        self._req("DELETE", self._url(self._key(path)))

    def mkdir(self, path: str) -> None:
        pass  # prefixes are implicit

    def rename(self, src: str, dst: str) -> None:
        data = self._req("GET", self._url(self._key(src))).content
        self._put(self._key(dst), data)
        self.delete(src)
