"""Worker process: block store + RPC + heartbeat + command execution.

Analog of /root/reference/curvine-worker/src/worker/worker_server.rs
(:46-298) and heartbeat_task.rs:27-60 (incremental block reports, async
execution of master commands) + worker/replication/
worker_replication_manager.rs:32-80 (semaphore-bounded block pushes) and
the load-task runner (curvine-data-transfer worker/task/
load_task_runner.rs:43).
"""
from __future__ import annotations

import asyncio
import logging
import random
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.conf import ClusterConf
from curvine_amd.model import (CMD_DELETE_BLOCK, CMD_REPLICATE, WorkerAddress,
                               WorkerInfo)
from curvine_amd.rpc.client import RpcClient
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.server import HandlerService, RpcServer
from curvine_amd.worker import registry
from curvine_amd.worker.block_store import BlockStore
from curvine_amd.worker.handlers import WorkerHandler

log = logging.getLogger("curvine.worker")


class WorkerService(HandlerService):
    def __init__(self, worker: "Worker"):
        self.worker = worker

    def get_message_handler(self):
        return WorkerHandler(self.worker)


class Worker:
    def __init__(self, conf: ClusterConf, worker_id: int | None = None,
                 device_id: int = -1):
        self.conf = conf
        self.worker_id = worker_id if worker_id is not None \
            else random.getrandbits(31)
        self.device_id = device_id
        self.store = BlockStore(conf.worker)
        self.rpc = RpcServer("worker", conf.worker.hostname,
                             conf.worker.rpc_port, WorkerService(self))
        self._hb_task: Optional[asyncio.Task] = None
        self._master: Optional[RpcClient] = None
        # ShortCircuitInfo pin leases: token -> (store reader, deadline)
        self.pins: dict[str, tuple] = {}
        self._stopped = asyncio.Event()
        self._repl_sem = asyncio.Semaphore(conf.worker.replication_concurrency)
        self._load_sem = asyncio.Semaphore(conf.job.worker_task_concurrency)
        self.decommissioning = False

    # ---------------- lifecycle ----------------
    async def start(self) -> "Worker":
        if self.conf.worker.native_data:
            try:
                from curvine_amd.worker.native_data import NativeDataFrontend
                self.rpc = NativeDataFrontend(
                    self, nthreads=self.conf.worker.data_threads)
            except Exception as e:  # noqa: BLE001 — asyncio fallback
                log.warning("native data frontend unavailable (%s); "
                            "using asyncio rpc server", e)
        await self.rpc.start()
        self.conf.worker.rpc_port = self.rpc.port
        registry.register(self.worker_id, self.store)
        await self._connect_master()
        await self._heartbeat_once(full_report=True)
        self._hb_task = asyncio.create_task(self._heartbeat_loop())
        log.info("worker %d started on %s:%d (device=%d, %d blocks)",
                 self.worker_id, self.conf.worker.hostname, self.rpc.port,
                 self.device_id, self.store.block_count())
        return self

    async def stop(self) -> None:
        self._stopped.set()
        if self._hb_task:
            self._hb_task.cancel()
            try:
                await self._hb_task
            except (asyncio.CancelledError, Exception):  # noqa: BLE001
                pass
        registry.unregister(self.worker_id)
        if self._master:
            await self._master.close()
        await self.rpc.stop()
        self.store.close()

    def address(self) -> WorkerAddress:
        return WorkerAddress(worker_id=self.worker_id,
                             hostname=self.conf.worker.hostname,
                             rpc_port=self.rpc.port,
                             device_id=self.device_id)

    def info(self) -> WorkerInfo:
        from curvine_amd.compat import component_info
        return WorkerInfo(address=self.address(),
                          storages=self.store.storages(),
                          component_info=component_info("worker"))

    # ---------------- heartbeats ----------------
    async def _connect_master(self) -> None:
        # ClusterConnector: follows the raft leader across masters
        from curvine_amd.rpc.client import ClusterConnector
        addrs = list(self.conf.client.master_addrs)
        own = f"{self.conf.master.hostname}:{self.conf.master.rpc_port}"
        if own not in addrs:
            addrs.append(own)
        if self._master is None:
            self._master = ClusterConnector(addrs,
                                            self.conf.client.rpc_timeout_ms)

    async def _heartbeat_loop(self) -> None:
        interval = self.conf.worker.heartbeat_interval_ms / 1000.0
        tick = 0
        while not self._stopped.is_set():
            try:
                await asyncio.sleep(interval)
                await self._heartbeat_once()
                tick += 1
                self.store.reap_deferred()
                now = __import__("time").monotonic()
                for tok in [t for t, (_, dl) in self.pins.items()
                            if dl < now]:
                    r, _ = self.pins.pop(tok)
                    try:
                        r.close()
                    except Exception:  # noqa: BLE001
                        pass
                if tick % 5 == 0:   # tier-pressure demotion sweep
                    loop = asyncio.get_event_loop()
                    await loop.run_in_executor(None,
                                               self.store.demote_coldest)
            except asyncio.CancelledError:
                return
            except Exception as e:  # noqa: BLE001
                log.warning("heartbeat failed: %s", e)
                try:
                    await self._connect_master()
                except Exception:  # noqa: BLE001
                    pass

    async def _heartbeat_once(self, full_report: bool = False) -> None:
        added, removed = self.store.take_deltas()
        if full_report:
            added = self.store.full_report()
        reply = await self._master.rpc(RpcCode.WorkerHeartbeat, {
            "worker": self.info().to_dict(),
            "added_blocks": added, "removed_blocks": removed})
        for cmd in reply.header.get("commands", []):
            asyncio.create_task(self._execute(cmd))

    async def _execute(self, cmd: dict) -> None:
        try:
            kind = cmd.get("cmd")
            if kind == CMD_DELETE_BLOCK:
                loop = asyncio.get_event_loop()
                await loop.run_in_executor(None, self.store.delete,
                                           cmd["block_id"])
            elif kind == CMD_REPLICATE:
                await self._replicate(cmd)
            elif kind == "load_task":
                await self._load_task(cmd)
            else:
                log.warning("unknown worker command %r", kind)
        except Exception as e:  # noqa: BLE001
            log.exception("command %s failed: %s", cmd.get("cmd"), e)

    # ---------------- replication (worker side) ----------------
    async def _replicate(self, cmd: dict) -> None:
        from curvine_amd.client.block_client import (BlockWriterRemote,
                                                     _native_data_lib,
                                                     _raise_wire_error)
        async with self._repl_sem:
            bid = cmd["block_id"]
            ok, error = True, ""
            try:
                loop = asyncio.get_event_loop()
                reader = await loop.run_in_executor(None,
                                                    self.store.open_reader, bid)
                try:
                    targets = [WorkerAddress.from_dict(t)
                               for t in cmd.get("targets", [])]
                    lib = _native_data_lib()
                    for t in targets:
                        if lib is not None:
                            # native push: one GIL-released streaming
                            # write from a host staging buffer
                            buf = bytearray(reader.length)
                            await loop.run_in_executor(
                                None, reader.read_into, 0, buf, 0,
                                reader.length)
                            status, hdr = await loop.run_in_executor(
                                None, lib.data_write_from, t.hostname,
                                t.rpc_port, bid, reader.length,
                                cmd.get("tier", ""), buf, 0, reader.length,
                                8 << 20, 4, False, reader.length)
                            if status == 5:
                                _raise_wire_error(hdr)
                            continue
                        w = BlockWriterRemote(t, bid, reader.length,
                                              cmd.get("tier", ""))
                        pos = 0
                        chunk = 4 << 20
                        while pos < reader.length:
                            n = min(chunk, reader.length - pos)
                            data = await loop.run_in_executor(
                                None, reader.read, pos, n)
                            await w.write(data)
                            pos += n
                        await w.commit(reader.length)
                finally:
                    reader.close()
            except Exception as e:  # noqa: BLE001
                ok, error = False, str(e)
                log.warning("replicate block %d failed: %s", bid, e)
            try:
                await self._master.rpc(RpcCode.ReportBlockReplicationResult, {
                    "block_id": bid, "job_id": cmd.get("job_id"),
                    "success": ok, "error": error})
            except Exception:  # noqa: BLE001
                pass

    # ---------------- load tasks (UFS -> cache ingest) ----------------
    async def _load_task(self, cmd: dict) -> None:
        from curvine_amd.client.filesystem import CurvineFileSystem
        from curvine_amd.ufs import get_ufs
        async with self._load_sem:
            ok, error = True, ""
            try:
                from curvine_amd.fault import fault_point
                fault_point("worker.load_task")
                ufs = get_ufs(cmd["ufs_path"], cmd.get("properties", {}))
                fs = CurvineFileSystem(self.conf)
                fs.client.local_worker_id = self.worker_id
                try:
                    writer = await fs.create(cmd["cv_path"], overwrite=True,
                                             replicas=cmd.get("replicas", 1))
                    loop = asyncio.get_event_loop()
                    reader = ufs.open(cmd["ufs_rel"])
                    chunk = self.conf.job.task_chunk_size
                    while True:
                        data = await loop.run_in_executor(None, reader.read, chunk)
                        if not data:
                            break
                        await writer.write(data)
                    reader.close()
                    await writer.complete()
                finally:
                    await fs.close()
            except Exception as e:  # noqa: BLE001
                ok, error = False, str(e)
                log.warning("load task %s failed: %s", cmd.get("task_id"), e)
            try:
                await self._master.rpc(RpcCode.ReportTask, {
                    "job_id": cmd.get("task_id", "").rsplit("-t", 1)[0],
                    "task_id": cmd.get("task_id"),
                    "success": ok, "error": error})
            except Exception:  # noqa: BLE001
                pass

    async def decommission(self) -> None:
        self.decommissioning = True
