"""FUSE daemon composition: asyncio client loop + channels + (optionally)
an embedded worker owning the local GPU's HBM arena.

The MI355X deployment model: one FUSE daemon per node (or per GPU),
embedding the worker so READs on HBM-cached blocks are in-process arena
reads.  `mount()` is the programmatic entry; `cv-fuse` CLI wraps it.
"""
from __future__ import annotations

import asyncio
import logging
import threading
from typing import Optional

from curvine_amd.client.filesystem import CurvineFileSystem
from curvine_amd.conf import ClusterConf
from curvine_amd.fuse.ops import CurvineFuseFs
from curvine_amd.fuse.session import FuseSession

log = logging.getLogger("curvine.fuse.daemon")


class FuseDaemon:
    def __init__(self, conf: ClusterConf, mnt_path: str | None = None,
                 embed_worker: bool = False, device_id: int = -1):
        self.conf = conf
        self.mnt_path = mnt_path or conf.fuse.mnt_path
        self.embed_worker = embed_worker
        self.device_id = device_id
        self.loop = asyncio.new_event_loop()
        self._loop_thread = threading.Thread(
            target=self._run_loop, daemon=True, name="curvine-fuse-loop")
        self.worker = None
        self.fs: Optional[CurvineFileSystem] = None
        self.fuse_fs: Optional[CurvineFuseFs] = None
        self.session: Optional[FuseSession] = None

    def _run_loop(self):
        asyncio.set_event_loop(self.loop)
        self.loop.run_forever()

    def call(self, coro, timeout: float = 120.0):
        return asyncio.run_coroutine_threadsafe(coro, self.loop).result(timeout)

    def start(self) -> "FuseDaemon":
        self._loop_thread.start()
        if self.embed_worker:
            from curvine_amd.worker.server import Worker

            async def mkworker():
                return await Worker(self.conf, device_id=self.device_id).start()
            self.worker = self.call(mkworker())

        async def mkfs():
            return CurvineFileSystem(self.conf)
        self.fs = self.call(mkfs())
        if self.worker is not None:
            self.fs.client.local_worker_id = self.worker.worker_id
        self.fuse_fs = CurvineFuseFs(self.fs, self.conf, self.loop)
        self.session = FuseSession(
            self.fuse_fs, self.mnt_path,
            channels=self.conf.fuse.mnt_number,
            max_write=self.conf.fuse.max_write).start()
        return self

    def stop(self) -> None:
        if self.session:
            self.session.stop()
        if self.fs:
            try:
                self.call(self.fs.close(), timeout=10)
            except Exception:  # noqa: BLE001
                pass
        if self.worker:
            try:
                self.call(self.worker.stop(), timeout=10)
            except Exception:  # noqa: BLE001
                pass
        self.loop.call_soon_threadsafe(self.loop.stop)
        self._loop_thread.join(timeout=5)
