"""fsspec filesystem for curvine ("cv://").

Analog of the reference's Python SDK
(/root/reference/curvine-libsdk/python/ `curvinefs` fsspec-style package +
curvine-libsdk-python pyo3 ABI python_abi.rs:12-88): a PyTorch DataLoader
or pandas/pyarrow reader pointed at ``cv://host:port/path`` works
unchanged.

    import fsspec
    fs = fsspec.filesystem("cv", master="127.0.0.1:8995")
    with fs.open("/data/shard-000.tar", "rb") as f: ...
"""
from __future__ import annotations

import io
from typing import Optional

try:
    from fsspec import AbstractFileSystem
    from fsspec.spec import AbstractBufferedFile
    _HAVE_FSSPEC = True
except ImportError:  # pragma: no cover
    _HAVE_FSSPEC = False

    class AbstractFileSystem:  # type: ignore[no-redef]
        def __init__(self, *a, **kw):
            pass

    class AbstractBufferedFile:  # type: ignore[no-redef]
        pass

from curvine_amd.client.filesystem import SyncFs
from curvine_amd.conf import ClusterConf


class CurvineFileSystemSpec(AbstractFileSystem):
    protocol = "cv"
    root_marker = "/"

    def __init__(self, master: str | None = None,
                 conf: ClusterConf | None = None, **kw):
        super().__init__(**kw)
        conf = conf or ClusterConf()
        if master:
            conf.client.master_addrs = [master]
        self.conf = conf
        self._fs: Optional[SyncFs] = None

    @property
    def fs(self) -> SyncFs:
        if self._fs is None:
            from curvine_amd.unified import UnifiedFileSystem

            sf = SyncFs.__new__(SyncFs)
            import asyncio
            import threading
            sf._own_loop = True
            sf.loop = asyncio.new_event_loop()
            sf._thread = threading.Thread(target=sf.loop.run_forever,
                                          daemon=True)
            sf._thread.start()

            async def mk():
                return UnifiedFileSystem(self.conf)
            sf.fs = sf.call(mk())
            self._fs = sf
        return self._fs

    @classmethod
    def _strip_protocol(cls, path):
        path = super()._strip_protocol(path) if _HAVE_FSSPEC else path
        if isinstance(path, str) and not path.startswith("/"):
            path = "/" + path
        return path or "/"

    # ---------------- metadata ----------------
    def _status_to_info(self, st) -> dict:
        # rooted names, matching _strip_protocol (pyarrow's dataset
        # discovery compares them against the normalized base dir)
        return {"name": st.path,
                "size": st.length,
                "type": "directory" if st.is_dir else "file",
                "mtime": st.mtime_ms / 1000.0}

    def info(self, path, **kw):
        st = self.fs.file_status(self._strip_protocol(path))
        return self._status_to_info(st)

    def ls(self, path, detail=True, **kw):
        infos = [self._status_to_info(s)
                 for s in self.fs.list_status(self._strip_protocol(path))]
        return infos if detail else [i["name"] for i in infos]

    def exists(self, path, **kw):
        return self.fs.exists(self._strip_protocol(path))

    def mkdir(self, path, create_parents=True, **kw):
        self.fs.mkdir(self._strip_protocol(path), create_parents=create_parents)

    def makedirs(self, path, exist_ok=True):
        self.fs.mkdir(self._strip_protocol(path), create_parents=True)

    def rmdir(self, path):
        self.fs.delete(self._strip_protocol(path), recursive=False)

    def _rm(self, path):
        self.fs.delete(self._strip_protocol(path), recursive=False)

    def rm(self, path, recursive=False, maxdepth=None):
        self.fs.delete(self._strip_protocol(path), recursive=recursive)

    def mv(self, src, dst, **kw):
        self.fs.rename(self._strip_protocol(src), self._strip_protocol(dst))

    def created(self, path):
        import datetime
        st = self.fs.file_status(self._strip_protocol(path))
        return datetime.datetime.fromtimestamp(st.mtime_ms / 1000.0)

    # ---------------- data ----------------
    def _open(self, path, mode="rb", block_size=None, **kw):
        return CurvineBufferedFile(self, self._strip_protocol(path), mode,
                                   block_size=block_size or 4 << 20, **kw)

    def cat_file(self, path, start=None, end=None, **kw):
        path = self._strip_protocol(path)
        r = self.fs.call(self.fs.fs.open(path))
        try:
            s = start or 0
            e = end if end is not None else r.length
            return self.fs.call(r.pread(s, max(0, e - s)))
        finally:
            r.close()


class CurvineBufferedFile(AbstractBufferedFile):
    def __init__(self, fs, path, mode, block_size=4 << 20, **kw):
        self._reader = None
        self._writer = None
        if _HAVE_FSSPEC:
            super().__init__(fs, path, mode, block_size=block_size, **kw)
        else:
            self.fs, self.path, self.mode = fs, path, mode
        if "r" in mode:
            self._reader = fs.fs.call(fs.fs.fs.open(path))
            if not _HAVE_FSSPEC:
                self.size = self._reader.length
        else:
            self._writer = fs.fs.call(
                fs.fs.fs.create(path, overwrite="w" in mode))

    def _fetch_range(self, start, end):
        return self.fs.fs.call(self._reader.pread(start, end - start))

    def _upload_chunk(self, final=False):
        data = self.buffer.getvalue()
        if data:
            self.fs.fs.call(self._writer.write(data))
        self.buffer = io.BytesIO()
        if final:
            self.fs.fs.call(self._writer.complete())
            self._writer = None
        return True

    def close(self):
        if getattr(self, "_reader", None) is not None:
            self._reader.close()
            self._reader = None
        if _HAVE_FSSPEC:
            super().close()
        elif self._writer is not None:
            self.fs.fs.call(self._writer.complete())
            self._writer = None


def register() -> None:
    """Register the cv:// protocol with fsspec."""
    if _HAVE_FSSPEC:
        import fsspec
        fsspec.register_implementation("cv", CurvineFileSystemSpec,
                                       clobber=True)


register()
