"""UnifiedFileSystem: mount-table-aware cache-or-UFS routing.

Analog of /root/reference/crates/client/curvine-unified-fs/src/unified/
unified_filesystem.rs (:107-143 routing, :496-626 open with cache-miss
fallback + bounded async-cache submit), fallback_fs_reader.rs (mid-read
failover to direct UFS) and write_cache_writer.rs (mirror writes to UFS),
mount_cache.rs (TTL'd mount table).
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.client.filesystem import CurvineFileSystem
from curvine_amd.client.reader import FsReader
from curvine_amd.conf import ClusterConf
from curvine_amd.model import FileStatus, FileType, MountInfo, now_ms
from curvine_amd.ufs import get_ufs

log = logging.getLogger("curvine.unified")


class MountCache:
    """TTL'd mount table (mount_cache.rs analog)."""

    def __init__(self, client, ttl_s: float = 10.0):
        self.client = client
        self.ttl = ttl_s
        self._table: list[MountInfo] = []
        self._at = 0.0

    async def table(self) -> list[MountInfo]:
        if time.monotonic() - self._at > self.ttl:
            try:
                self._table = await self.client.get_mount_table()
                self._at = time.monotonic()
            except err.FsError:
                pass
        return self._table

    async def lookup(self, path: str) -> Optional[MountInfo]:
        best = None
        for mi in await self.table():
            p = mi.curvine_path
            if path == p or path.startswith(p + "/"):
                if best is None or len(p) > len(best.curvine_path):
                    best = mi
        return best

    def invalidate(self) -> None:
        self._at = 0.0


class FallbackUfsReader:
    """Reader over the UFS directly; same surface as FsReader."""

    def __init__(self, ufs, rel: str, length: int):
        self.ufs = ufs
        self.rel = rel
        self.length = length
        self.pos = 0

    async def pread_into(self, off: int, out, out_off: int, n: int) -> int:
        n = max(0, min(n, self.length - off))
        if n == 0:
            return 0
        loop = asyncio.get_event_loop()

        def do_read():
            r = self.ufs.open(self.rel, off)
            got = 0
            try:
                while got < n:
                    chunk = r.read(min(n - got, 4 << 20))
                    if not chunk:
                        break
                    out[out_off + got:out_off + got + len(chunk)] = chunk
                    got += len(chunk)
            finally:
                r.close()
            return got
        return await loop.run_in_executor(None, do_read)

    async def pread(self, off: int, n: int) -> bytes:
        out = bytearray(max(0, min(n, self.length - off)))
        got = await self.pread_into(off, out, 0, len(out))
        return bytes(out[:got])

    async def read(self, n: int = -1) -> bytes:
        if n < 0:
            n = self.length - self.pos
        data = await self.pread(self.pos, n)
        self.pos += len(data)
        return data

    def seek(self, pos: int) -> None:
        self.pos = pos

    def close(self) -> None:
        pass


class FallbackFsReader:
    """Cache reader that fails over to UFS mid-read on worker errors
    (fallback_fs_reader.rs:21-55 analog).  The UFS reader is constructed
    lazily (a coroutine factory) so healthy cache reads never touch the
    UFS."""

    def __init__(self, cache_reader: FsReader, ufs_factory):
        self.cache = cache_reader
        self._ufs_factory = ufs_factory
        self.ufs: Optional[FallbackUfsReader] = None
        self.length = cache_reader.length
        self.pos = 0
        self._failed = False

    async def _ufs_reader(self) -> FallbackUfsReader:
        if self.ufs is None:
            self.ufs = await self._ufs_factory()
        return self.ufs

    async def pread_into(self, off: int, out, out_off: int, n: int) -> int:
        if not self._failed:
            try:
                got = await self.cache.pread_into(off, out, out_off, n)
                if got > 0:
                    return got
            except Exception as e:  # noqa: BLE001
                log.warning("cache read failed mid-stream, UFS fallback: %s", e)
                self._failed = True
        ufs = await self._ufs_reader()
        return await ufs.pread_into(off, out, out_off, n)

    async def pread(self, off: int, n: int) -> bytes:
        out = bytearray(max(0, min(n, self.length - off)))
        got = await self.pread_into(off, out, 0, len(out))
        return bytes(out[:got])

    async def read(self, n: int = -1) -> bytes:
        if n < 0:
            n = self.length - self.pos
        data = await self.pread(self.pos, n)
        self.pos += len(data)
        return data

    def seek(self, pos: int) -> None:
        self.pos = pos

    def close(self) -> None:
        self.cache.close()
        if self.ufs is not None:
            self.ufs.close()


class WriteCacheWriter:
    """Writes to the cache AND mirrors to the UFS
    (write_cache_writer.rs:49-264 analog; mirror is synchronous-at-close)."""

    def __init__(self, cache_writer, ufs, rel: str):
        self.cache = cache_writer
        self.ufs = ufs
        self.rel = rel
        self._uw = None
        self.status = cache_writer.status

    async def write(self, data) -> int:
        loop = asyncio.get_event_loop()
        if self._uw is None:
            self._uw = await loop.run_in_executor(None, self.ufs.create, self.rel)
        n = await self.cache.write(data)
        payload = bytes(data)
        await loop.run_in_executor(None, self._uw.write, payload)
        return n

    async def flush(self) -> None:
        await self.cache.flush()

    async def complete(self) -> FileStatus:
        loop = asyncio.get_event_loop()
        if self._uw is None:
            self._uw = await loop.run_in_executor(None, self.ufs.create, self.rel)
        await loop.run_in_executor(None, self._uw.close)
        return await self.cache.complete()

    async def abort(self) -> None:
        await self.cache.abort()


async def _ready(x):
    return x


def _ufs_status(path: str, info: dict) -> FileStatus:
    return FileStatus(
        inode_id=0, path=path,
        name=path.rsplit("/", 1)[-1],
        file_type=int(FileType.DIR) if info.get("is_dir") else int(FileType.FILE),
        length=info.get("length", 0), is_complete=True,
        mtime_ms=info.get("mtime_ms", now_ms()), atime_ms=now_ms())


class UnifiedFileSystem(CurvineFileSystem):
    """CurvineFileSystem + UFS fallthrough under mount points."""

    def __init__(self, conf: ClusterConf | None = None):
        super().__init__(conf)
        self.mounts = MountCache(self.client)
        self._cache_inflight = 0
        self._cache_pending: set[str] = set()

    async def mount(self, curvine_path, ufs_path, properties=None,
                    cache_mode="cache", auto_cache=True):
        out = await super().mount(curvine_path, ufs_path, properties,
                                  cache_mode, auto_cache)
        self.mounts.invalidate()   # own mutations must be visible at once
        return out

    async def unmount(self, curvine_path):
        out = await super().unmount(curvine_path)
        self.mounts.invalidate()
        return out

    async def _route(self, path: str):
        """(mount_info, ufs, rel) or (None, None, None)."""
        mi = await self.mounts.lookup(path)
        if mi is None:
            return None, None, None
        rel = path[len(mi.curvine_path):] or "/"
        loop = asyncio.get_event_loop()
        ufs = await loop.run_in_executor(None, get_ufs, mi.ufs_path,
                                         mi.properties)
        return mi, ufs, rel

    # ---------------- reads ----------------
    async def open(self, path: str):
        cached = None
        healthy = False
        try:
            fb = await self.client.open(path)
            cached = FsReader(self.client, fb)
            healthy = fb.status.is_complete and \
                (fb.blocks or fb.status.length == 0) and \
                all(b.locations for b in fb.blocks)
        except err.FileNotFound:
            pass
        mi, ufs, rel = await self._route(path)
        if mi is None:
            if cached is not None:
                return cached
            raise err.FileNotFound(path)

        loop = asyncio.get_event_loop()

        async def make_ufs_reader() -> FallbackUfsReader:
            info = await loop.run_in_executor(None, ufs.status, rel)
            if info is None:
                raise err.FileNotFound(path)
            return FallbackUfsReader(ufs, rel, info["length"])

        if cached is not None and healthy:
            # healthy cache read, but keep a lazy UFS failover behind it
            return FallbackFsReader(cached, make_ufs_reader)
        ufs_reader = await make_ufs_reader()
        if mi.auto_cache and self.conf.client.auto_cache:
            await self._submit_auto_cache(path)
        if cached is not None and cached.length == ufs_reader.length:
            return FallbackFsReader(cached, lambda: _ready(ufs_reader))
        return ufs_reader

    async def _submit_auto_cache(self, path: str) -> None:
        """Bounded async-cache submit (unified_filesystem.rs:60-79)."""
        if path in self._cache_pending or \
                self._cache_inflight >= self.conf.client.auto_cache_max_inflight:
            return
        self._cache_pending.add(path)
        self._cache_inflight += 1

        async def submit():
            try:
                await self.client.submit_job(path, recursive=False)
            except err.FsError as e:
                log.debug("auto-cache submit %s: %s", path, e)
            finally:
                self._cache_inflight -= 1
        asyncio.create_task(submit())

    # ---------------- metadata with UFS fallthrough ----------------
    async def file_status(self, path: str) -> FileStatus:
        try:
            return await self.client.file_status(path)
        except err.FileNotFound:
            mi, ufs, rel = await self._route(path)
            if mi is None:
                raise
            loop = asyncio.get_event_loop()
            info = await loop.run_in_executor(None, ufs.status, rel)
            if info is None:
                raise
            return _ufs_status(path, info)

    async def exists(self, path: str) -> bool:
        if await self.client.exists(path):
            return True
        mi, ufs, rel = await self._route(path)
        if mi is None:
            return False
        loop = asyncio.get_event_loop()
        return await loop.run_in_executor(None, ufs.status, rel) is not None

    async def list_status(self, path: str) -> list[FileStatus]:
        cached: dict[str, FileStatus] = {}
        try:
            for s in await self.client.list_status(path):
                cached[s.name] = s
        except err.FileNotFound:
            pass
        mi, ufs, rel = await self._route(path)
        if mi is not None:
            loop = asyncio.get_event_loop()
            try:
                entries = await loop.run_in_executor(
                    None, lambda: ufs.list_files(rel, recursive=False))
                base = path.rstrip("/")
                for e in entries:
                    name = e["path"].rstrip("/").rsplit("/", 1)[-1]
                    if name not in cached:
                        cached[name] = _ufs_status(f"{base}/{name}", e)
            except err.FsError:
                pass
        if not cached and mi is None and not await self.client.exists(path):
            raise err.FileNotFound(path)
        return sorted(cached.values(), key=lambda s: s.name)

    # ---------------- writes ----------------
    async def create(self, path: str, overwrite: bool = False, **kw):
        writer = await super().create(path, overwrite=overwrite, **kw)
        mi, ufs, rel = await self._route(path)
        if mi is not None and mi.cache_mode == "fs":
            return WriteCacheWriter(writer, ufs, rel)
        return writer

    async def delete(self, path: str, recursive: bool = False) -> int:
        n = 0
        try:
            n = await self.client.delete(path, recursive)
        except err.FileNotFound:
            pass
        mi, ufs, rel = await self._route(path)
        if mi is not None and mi.cache_mode == "fs":
            loop = asyncio.get_event_loop()
            await loop.run_in_executor(None, ufs.delete, rel, recursive)
        return n

    async def read_all(self, path: str) -> bytes:
        r = await self.open(path)
        try:
            return await r.pread(0, r.length)
        finally:
            r.close()
