"""CurvineFileSystem — the client facade.

Analog of /root/reference/crates/client/curvine-client-core/src/file/
curvine_filesystem.rs:42 (+ FsContext fs_context.rs:48-225).  Async API;
`SyncFs` wraps it in a dedicated event-loop thread for synchronous callers
(CLI, Python SDK, benchmarks).
"""
from __future__ import annotations

import asyncio
import threading

from curvine_amd import errors as err
from curvine_amd.client.fs_client import FsClient
from curvine_amd.client.reader import FsReader
from curvine_amd.client.writer import FsWriter
from curvine_amd.conf import ClusterConf
from curvine_amd.model import FileStatus


class CurvineFileSystem:
    def __init__(self, conf: ClusterConf | None = None):
        self.conf = conf or ClusterConf()
        self.client = FsClient(self.conf)

    # namespace passthrough
    async def mkdir(self, path: str, mode: int = 0o755,
                    create_parents: bool = True) -> FileStatus:
        return await self.client.mkdir(path, mode, create_parents)

    async def delete(self, path: str, recursive: bool = False) -> int:
        return await self.client.delete(path, recursive)

    async def rename(self, src: str, dst: str) -> None:
        await self.client.rename(src, dst)

    async def file_status(self, path: str) -> FileStatus:
        return await self.client.file_status(path)

    async def exists(self, path: str) -> bool:
        return await self.client.exists(path)

    async def list_status(self, path: str) -> list[FileStatus]:
        return await self.client.list_status(path)

    async def set_attr(self, path: str, **attrs) -> FileStatus:
        return await self.client.set_attr(path, **attrs)

    async def symlink(self, path: str, target: str) -> FileStatus:
        return await self.client.symlink(path, target)

    async def link(self, src: str, dst: str) -> FileStatus:
        return await self.client.link(src, dst)

    async def resize(self, path: str, length: int) -> FileStatus:
        return await self.client.resize(path, length)

    async def free(self, path: str, recursive: bool = False) -> int:
        return await self.client.free(path, recursive)

    async def get_master_info(self) -> dict:
        return await self.client.get_master_info()

    # data path
    async def create(self, path: str, overwrite: bool = False,
                     replicas: int = 0, block_size: int = 0,
                     storage_tier: str = "") -> FsWriter:
        st = await self.client.create(path, overwrite, replicas, block_size,
                                      storage_tier)
        return FsWriter(self.client, st)

    async def append(self, path: str) -> FsWriter:
        fb = await self.client.append(path)
        w = FsWriter(self.client, fb.status)
        w._block_lens = [b.block.length for b in fb.blocks]
        w.pos = fb.status.length
        return w

    async def open(self, path: str) -> FsReader:
        fb = await self.client.open(path)
        return FsReader(self.client, fb)

    async def write_files_batch(self, files: dict, overwrite: bool = True,
                                storage_tier: str = "") -> list[FileStatus]:
        """Small-file fast path (BatchBlockWriter analog,
        /root/reference/.../block/batch_block_writer.rs +
        batch_write_handler.rs:32-150): one CreateFilesBatch, one
        AddBlocksBatch, payloads grouped per worker into WriteBlocksBatch
        frames, one CompleteFilesBatch."""
        from curvine_amd.rpc.codes import RpcCode
        from curvine_amd.rpc.message import MAX_DATA_SIZE
        paths = list(files.keys())
        tier = storage_tier or self.conf.client.storage_tier
        reply = await self.client.connector.rpc(RpcCode.CreateFilesBatch, {
            "files": [{"path": p, "overwrite": overwrite,
                       "storage_tier": tier,
                       "block_size": self.conf.client.block_size}
                      for p in paths]})
        reply = await self.client.connector.rpc(RpcCode.AddBlocksBatch, {
            "blocks": [{"path": p} for p in paths],
            "client_host": self.client.client_host,
            "client_worker_id": self.client.local_worker_id})
        from curvine_amd.model import LocatedBlock
        lbs = [LocatedBlock.from_dict(b) for b in reply.header["blocks"]]
        # group by (first) target worker
        groups: dict = {}
        for p, lb in zip(paths, lbs):
            addr = lb.locations[0]
            groups.setdefault(addr.key(), (addr, []))[1].append((p, lb))
        from curvine_amd.client.block_client import factory
        from curvine_amd.worker import registry
        commits: dict[str, dict] = {}
        for addr, items in groups.values():
            store = registry.lookup(addr.worker_id)
            if store is not None:
                loop = asyncio.get_event_loop()
                for p, lb in items:
                    data = files[p]

                    def write_one(bid=lb.block.block_id, data=data):
                        w = store.create_writer(bid, max(len(data), 1), tier)
                        if data:
                            w.write(data)
                        return store.finalize(bid, len(data))
                    t = await loop.run_in_executor(None, write_one)
                    commits[p] = {"block_id": lb.block.block_id,
                                  "locations": [addr.worker_id], "tiers": [t]}
                continue
            client = await factory().get(addr.hostname, addr.rpc_port)
            batch, payload, size = [], [], 0
            pend = list(items)
            while pend:
                p, lb = pend.pop(0)
                data = files[p]
                if size + len(data) > MAX_DATA_SIZE and batch:
                    await self._flush_batch(client, batch, payload, commits,
                                            addr, tier)
                    batch, payload, size = [], [], 0
                batch.append((p, lb))
                payload.append(data)
                size += len(data)
            if batch:
                await self._flush_batch(client, batch, payload, commits,
                                        addr, tier)
        reply = await self.client.connector.rpc(RpcCode.CompleteFilesBatch, {
            "files": [{"path": p, "length": len(files[p]),
                       "block_lens": [len(files[p])] if files[p] else [],
                       "commits": [commits[p]] if p in commits else []}
                      for p in paths]})
        return [FileStatus.from_dict(s) for s in reply.header["statuses"]]

    async def _flush_batch(self, client, batch, payload, commits, addr, tier):
        from curvine_amd.rpc.codes import RpcCode
        blocks = [{"block_id": lb.block.block_id, "length": len(d),
                   "tier": tier}
                  for (p, lb), d in zip(batch, payload)]
        reply = await client.rpc(RpcCode.WriteBlocksBatch,
                                 {"blocks": blocks}, b"".join(payload))
        for p, lb in batch:
            commits[p] = {"block_id": lb.block.block_id,
                          "locations": [addr.worker_id], "tiers": [tier]}

    async def write_all(self, path: str, data, overwrite: bool = True,
                        **kw) -> FileStatus:
        w = await self.create(path, overwrite=overwrite, **kw)
        await w.write(data)
        return await w.complete()

    async def read_all(self, path: str) -> bytes:
        r = await self.open(path)
        try:
            return await r.pread(0, r.length)
        finally:
            r.close()

    async def close(self) -> None:
        await self.client.close()

    # mounts / jobs passthrough
    async def mount(self, curvine_path, ufs_path, properties=None,
                    cache_mode="cache", auto_cache=True):
        return await self.client.mount(curvine_path, ufs_path, properties,
                                       cache_mode, auto_cache)

    async def unmount(self, curvine_path):
        return await self.client.unmount(curvine_path)

    async def get_mount_table(self):
        return await self.client.get_mount_table()

    async def submit_job(self, path, recursive=True, replicas=1):
        return await self.client.submit_job(path, recursive, replicas)

    async def job_status(self, job_id):
        return await self.client.job_status(job_id)


class SyncFs:
    """Synchronous wrapper: runs a private event loop in a daemon thread and
    proxies coroutine calls (the analog of the reference's sync_client)."""

    def __init__(self, conf: ClusterConf | None = None,
                 loop: asyncio.AbstractEventLoop | None = None):
        self._own_loop = loop is None
        if loop is None:
            self.loop = asyncio.new_event_loop()
            self._thread = threading.Thread(target=self.loop.run_forever,
                                            daemon=True, name="curvine-syncfs")
            self._thread.start()
        else:
            self.loop = loop
            self._thread = None
        self.fs = self.call(self._make_fs(conf))

    @staticmethod
    async def _make_fs(conf):
        return CurvineFileSystem(conf)

    def call(self, coro, timeout: float = 300.0):
        fut = asyncio.run_coroutine_threadsafe(coro, self.loop)
        return fut.result(timeout)

    def __getattr__(self, name):
        target = getattr(self.fs, name)
        if asyncio.iscoroutinefunction(target):
            def proxy(*a, **kw):
                return self.call(target(*a, **kw))
            return proxy
        return target

    def read_file(self, path: str) -> bytes:
        return self.call(self.fs.read_all(path))

    def pread(self, path: str, start: int, length: int | None = None) -> bytes:
        async def go():
            r = await self.fs.open(path)
            try:
                n = (r.length - start) if length is None else length
                return await r.pread(start, max(0, n))
            finally:
                r.close()
        return self.call(go())

    def open_writer(self, path: str, overwrite: bool = True):
        """Sync streaming writer: .write(bytes) / .close() -> FileStatus."""
        w = self.call(self.fs.create(path, overwrite=overwrite))
        sync = self

        class _W:
            def write(self, data):
                return sync.call(w.write(data))

            def close(self):
                return sync.call(w.complete())

            def abort(self):
                return sync.call(w.abort())

            def __enter__(self):
                return self

            def __exit__(self, et, ev, tb):
                if et is None:
                    self.close()
                else:
                    self.abort()
        return _W()

    def write_file(self, path: str, data, **kw) -> FileStatus:
        return self.call(self.fs.write_all(path, data, **kw))

    def shutdown(self) -> None:
        try:
            self.call(self.fs.close(), timeout=10)
        except Exception:  # noqa: BLE001
            pass
        if self._own_loop:
            self.loop.call_soon_threadsafe(self.loop.stop)
            if self._thread:
                self._thread.join(timeout=5)
