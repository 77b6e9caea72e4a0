"""Under-filesystem (UFS) adapters.

Analog of the reference's `curvine-ufs-api` + opendal adapter
(/root/reference/crates/common/curvine-ufs-api,
crates/adapters/curvine-ufs-opendal): a minimal chunk-reader/writer
abstraction over external storage.  Schemes:

* ``file://`` / bare paths — local directory (testing + node-local NVMe)
* ``s3://`` — S3 object storage via raw HTTP (requests); functional when an
  endpoint is reachable, unit-tested against the local adapter since this
  environment has no network.
"""
from __future__ import annotations

from curvine_amd import errors as err
from curvine_amd.ufs.base import UnderFs  # noqa: F401
from curvine_amd.ufs.local import LocalUfs
from curvine_amd.ufs.s3 import S3Ufs

_CACHE: dict[tuple, UnderFs] = {}


def get_ufs(uri: str, properties: dict | None = None) -> UnderFs:
    properties = properties or {}
    key = (uri, tuple(sorted(properties.items())))
    if key in _CACHE:
        return _CACHE[key]
    if uri.startswith("s3://") or uri.startswith("oss://"):
        fs: UnderFs = S3Ufs(uri, properties)
    elif uri.startswith("gs://"):
        # GCS speaks the S3 XML protocol in interoperability mode (HMAC
        # keys sign with SigV4 exactly like S3); default the endpoint
        fs = S3Ufs(uri, {"endpoint": "https://storage.googleapis.com",
                         **properties})
    elif uri.startswith("az://") or uri.startswith("abfs://"):
        from curvine_amd.ufs.azure import AzureUfs
        fs = AzureUfs(uri, properties)
    elif uri.startswith("hdfs://") or uri.startswith("webhdfs://"):
        from curvine_amd.ufs.webhdfs import WebHdfsUfs
        fs = WebHdfsUfs(uri, properties)
    elif uri.startswith("file://"):
        fs = LocalUfs(uri[len("file://"):])
    elif uri.startswith("/"):
        fs = LocalUfs(uri)
    else:
        raise err.Unsupported(f"ufs scheme of {uri!r}")
    _CACHE[key] = fs
    return fs
