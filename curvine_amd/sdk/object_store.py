"""Object-store facade over the cache (curvine-lancedb analog).

The reference's `curvine-lancedb` crate adapts Curvine to the Rust
`object_store` trait so LanceDB tables live on cv:// storage
(/root/reference/curvine-lancedb/src/object_store.rs:406-420 put modes,
safe_commit.rs:25-76 conditional-put commit handler).  The `lancedb`
package does not exist in this environment, so the deliverable here is
the store surface itself — the same operations and commit semantics any
table format (Lance, Delta-style logs, custom manifests) needs:

* ``put(mode="create")`` — atomic create-if-absent (the master's
  CreateFile with overwrite=False is the linearization point; two
  racing committers: exactly one wins)
* ``put(mode="overwrite")``, ``get`` (with range), ``head``, ``delete``,
  ``list``/``list_with_delimiter``, ``copy``/``copy_if_not_exists``,
  ``rename`` (atomic at the master)
* ``ConditionalPutCommitter`` — the ConditionalPutCommitHandler analog:
  versioned manifest commits where version N's writer must lose if N
  already exists

Arrow/parquet datasets work through the fsspec adapter (`cv://`,
sdk/fsspec_fs.py) — see tests/test_sdk.py::test_pyarrow_parquet_dataset.
"""
from __future__ import annotations

from dataclasses import dataclass
from typing import Optional

from curvine_amd import errors as err


@dataclass
class ObjectMeta:
    key: str
    size: int
    mtime_ms: int


class CurvineObjectStore:
    """Synchronous object-store surface over a SyncFs (or per-call fs)."""

    def __init__(self, sync_fs, prefix: str = "/"):
        self.fs = sync_fs
        self.prefix = "/" + prefix.strip("/")

    def _p(self, key: str) -> str:
        base = "" if self.prefix == "/" else self.prefix
        return f"{base}/{key.lstrip('/')}"

    # ---------------- puts ----------------
    def put(self, key: str, data: bytes, mode: str = "overwrite") -> None:
        if mode == "create":
            # atomic create-if-absent: the master rejects the loser with
            # FileAlreadyExists before any byte is written
            self.fs.write_file(self._p(key), data, overwrite=False)
        elif mode == "overwrite":
            self.fs.write_file(self._p(key), data, overwrite=True)
        else:
            raise err.InvalidArgument(f"put mode {mode!r}")

    def put_multipart(self, key: str):
        """Streaming writer for large objects (multipart analog: one
        logical stream; block granularity handles the sharding)."""
        return self.fs.open_writer(self._p(key), overwrite=True)

    # ---------------- reads ----------------
    def get(self, key: str, start: int = 0,
            length: Optional[int] = None) -> bytes:
        p = self._p(key)
        if start == 0 and length is None:
            return self.fs.read_file(p)
        return self.fs.pread(p, start, length)

    def head(self, key: str) -> ObjectMeta:
        st = self.fs.file_status(self._p(key))
        return ObjectMeta(key=key, size=st.length, mtime_ms=st.mtime_ms)

    def exists(self, key: str) -> bool:
        return self.fs.exists(self._p(key))

    # ---------------- namespace ----------------
    def delete(self, key: str) -> None:
        self.fs.delete(self._p(key))

    def list(self, prefix: str = "") -> list[ObjectMeta]:
        """All objects under prefix, recursively, sorted by key."""
        out: list[ObjectMeta] = []
        base = self._p(prefix) if prefix else self.prefix
        strip = ("" if self.prefix == "/" else self.prefix) + "/"

        def walk(path):
            try:
                sts = self.fs.list_status(path)
            except err.FileNotFound:
                return
            for s in sts:
                if s.is_dir:
                    walk(s.path)
                else:
                    out.append(ObjectMeta(key=s.path[len(strip):],
                                          size=s.length, mtime_ms=s.mtime_ms))
        walk(base)
        out.sort(key=lambda m: m.key)
        return out

    def list_with_delimiter(self, prefix: str = "") -> tuple[list[str],
                                                             list[ObjectMeta]]:
        """One level: (common_prefixes, objects)."""
        base = self._p(prefix) if prefix else self.prefix
        strip = ("" if self.prefix == "/" else self.prefix) + "/"
        dirs, objs = [], []
        try:
            sts = self.fs.list_status(base)
        except err.FileNotFound:
            return [], []
        for s in sts:
            if s.is_dir:
                dirs.append(s.path[len(strip):])
            else:
                objs.append(ObjectMeta(key=s.path[len(strip):],
                                       size=s.length, mtime_ms=s.mtime_ms))
        return sorted(dirs), sorted(objs, key=lambda m: m.key)

    def copy(self, src: str, dst: str, overwrite: bool = True) -> None:
        if not overwrite and self.exists(dst):
            raise err.FileAlreadyExists(dst)
        self.put(dst, self.get(src), mode="overwrite")

    def copy_if_not_exists(self, src: str, dst: str) -> None:
        # atomic: the create of dst is the linearization point
        self.put(dst, self.get(src), mode="create")

    def rename(self, src: str, dst: str) -> None:
        """Atomic at the master (single journal entry)."""
        self.fs.rename(self._p(src), self._p(dst))


class ConditionalPutCommitter:
    """ConditionalPutCommitHandler analog (safe_commit.rs:34-36): commit
    version N by atomically creating its manifest; a racing committer of
    the same N loses with ``CommitConflict``."""

    def __init__(self, store: CurvineObjectStore, table_prefix: str):
        self.store = store
        self.prefix = table_prefix.strip("/")

    def manifest_key(self, version: int) -> str:
        return f"{self.prefix}/_versions/{version:020d}.manifest"

    def latest_version(self) -> int:
        ms = self.store.list(f"{self.prefix}/_versions")
        return max((int(m.key.rsplit("/", 1)[-1].split(".")[0]) for m in ms),
                   default=0)

    def commit(self, version: int, manifest: bytes) -> None:
        try:
            self.store.put(self.manifest_key(version), manifest, mode="create")
        except err.FileAlreadyExists:
            raise CommitConflict(version)

    def read_manifest(self, version: int) -> bytes:
        return self.store.get(self.manifest_key(version))


class CommitConflict(Exception):
    def __init__(self, version: int):
        super().__init__(f"version {version} already committed")
        self.version = version
