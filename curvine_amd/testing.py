"""MiniCluster: real master + workers in one process for tests/benches.

Analog of the reference's MiniCluster
(/root/reference/curvine-server/src/test/mini_cluster.rs:38-146) and the
`Testing` helpers (curvine-tests/src/testing.rs): every server binds port 0
(ephemeral) so parallel test processes never collide.
"""
from __future__ import annotations

import asyncio
import os
import tempfile
from typing import Optional

from curvine_amd.conf import ClusterConf
from curvine_amd.client.filesystem import CurvineFileSystem
from curvine_amd.master.server import Master
from curvine_amd.worker.server import Worker


def test_conf(tmp_dir: str, data_dirs: list[str] | None = None) -> ClusterConf:
    conf = ClusterConf()
    conf.testing = True
    conf.master.rpc_port = 0
    conf.worker.rpc_port = 0
    conf.master.heartbeat_check_ms = 200
    conf.master.worker_expire_ms = 5_000
    conf.worker.heartbeat_interval_ms = 100
    conf.journal.journal_dir = os.path.join(tmp_dir, "journal")
    conf.worker.data_dirs = data_dirs or [
        f"[MEM:64MB]{tmp_dir}/mem",
        f"[SSD:1GB]{tmp_dir}/ssd",
    ]
    conf.master.block_size = 4 << 20
    conf.client.block_size = 4 << 20
    conf.client.write_chunk_size = 256 << 10
    conf.client.read_chunk_size = 256 << 10
    return conf


class MiniCluster:
    def __init__(self, conf: ClusterConf | None = None, workers: int = 1,
                 tmp_dir: str | None = None,
                 worker_dirs: list[list[str]] | None = None):
        self._tmp = None
        if conf is None:
            if tmp_dir is None:
                self._tmp = tempfile.TemporaryDirectory(prefix="curvine-test-")
                tmp_dir = self._tmp.name
            conf = test_conf(tmp_dir)
        self.conf = conf
        self.tmp_dir = tmp_dir
        self.n_workers = workers
        self.worker_dirs = worker_dirs
        self.master: Optional[Master] = None
        self.workers: list[Worker] = []

    async def start(self) -> "MiniCluster":
        self.master = await Master(self.conf).start()
        # point workers/clients at the actual ephemeral port
        addr = f"{self.conf.master.hostname}:{self.master.rpc.port}"
        self.conf.client.master_addrs = [addr]
        for i in range(self.n_workers):
            wconf = test_conf(self.tmp_dir) if self._tmp else self.conf
            import copy
            wc = copy.deepcopy(self.conf)
            wc.worker.rpc_port = 0
            if self.worker_dirs and i < len(self.worker_dirs):
                wc.worker.data_dirs = self.worker_dirs[i]
            elif self.n_workers == 1:
                pass   # single worker: honour conf.worker.data_dirs as given
            else:
                wc.worker.data_dirs = [
                    f"[MEM:64MB]{self.tmp_dir}/w{i}/mem",
                    f"[SSD:1GB]{self.tmp_dir}/w{i}/ssd",
                ]
            w = await Worker(wc, worker_id=i + 1).start()
            self.workers.append(w)
        return self

    def client_conf(self) -> ClusterConf:
        import copy
        c = copy.deepcopy(self.conf)
        c.client.master_addrs = [
            f"{self.conf.master.hostname}:{self.master.rpc.port}"]
        return c

    def fs(self) -> CurvineFileSystem:
        return CurvineFileSystem(self.client_conf())

    async def stop(self) -> None:
        for w in self.workers:
            await w.stop()
        self.workers = []
        if self.master:
            await self.master.stop()
            self.master = None
        if self._tmp:
            self._tmp.cleanup()
            self._tmp = None

    async def __aenter__(self) -> "MiniCluster":
        return await self.start()

    async def __aexit__(self, *a) -> None:
        await self.stop()


class SyncMiniCluster:
    """MiniCluster running on a dedicated event-loop thread, for callers
    that are not themselves async (FUSE tests, CLI tests, benchmarks)."""

    def __init__(self, **kw):
        import threading
        self.loop = asyncio.new_event_loop()
        self._thread = threading.Thread(target=self._run, daemon=True,
                                        name="curvine-minicluster")
        self._thread.start()
        self.mc = MiniCluster(**kw)

    def _run(self):
        asyncio.set_event_loop(self.loop)
        self.loop.run_forever()

    def call(self, coro, timeout: float = 120.0):
        return asyncio.run_coroutine_threadsafe(coro, self.loop).result(timeout)

    def start(self) -> "SyncMiniCluster":
        self.call(self.mc.start())
        return self

    def stop(self) -> None:
        try:
            self.call(self.mc.stop(), timeout=30)
        finally:
            self.loop.call_soon_threadsafe(self.loop.stop)
            self._thread.join(timeout=5)

    def client_conf(self) -> ClusterConf:
        return self.mc.client_conf()

    @property
    def workers(self):
        return self.mc.workers

    @property
    def master(self):
        return self.mc.master
