"""Standalone data-transfer service.

Analog of /root/reference/crates/server/curvine-data-transfer/ (store
backends transfer/mod.rs:1-27, planner, scheduler, TransferService RPC
codes 46-54): the load/transfer orchestrator as its OWN process with its
own durable store, talking to the master purely over RPC — mount lookup
(GetMountTable), worker discovery (GetFilesystemInfo), task dispatch
(SubmitTask -> worker heartbeat commands) and task reports forwarded by
the master.  The same job logic also runs embedded in the master (the
reference supports both deployments); `conf.job.service_addr` selects.

Store backends: memory and sqlite (tested).  The reference also ships
mysql/postgres stores; this image has neither the servers nor client
drivers (no pymysql/psycopg, no network), so those backends are not
implemented rather than shipped untestable.
"""
from __future__ import annotations

import asyncio
import logging
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.conf import ClusterConf
from curvine_amd.master.jobs import JobManager
from curvine_amd.model import MountInfo
from curvine_amd.rpc.client import ClusterConnector
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.message import Message
from curvine_amd.rpc.server import HandlerService, RpcServer

log = logging.getLogger("curvine.transfer")


class RemoteEnv:
    """JobManager's view of the cluster when running OUTSIDE the master:
    everything goes over RPC (the reference's ClusterMetadataCache +
    master client)."""

    def __init__(self, conf: ClusterConf):
        self.conf = conf
        self.conn = ClusterConnector(list(conf.client.master_addrs),
                                     conf.client.rpc_timeout_ms)
        self._loop: Optional[asyncio.AbstractEventLoop] = None

    def _call(self, coro):
        return asyncio.get_event_loop().run_until_complete(coro) \
            if not asyncio.get_event_loop().is_running() else None

    async def mounts_lookup(self, path: str):
        r = await self.conn.rpc(RpcCode.GetMountTable, {})
        best = None
        for m in r.header.get("mounts", []):
            cp = m.get("curvine_path", "")
            if path == cp or path.startswith(cp.rstrip("/") + "/"):
                if best is None or len(cp) > len(best["curvine_path"]):
                    best = m
        return MountInfo.from_dict(best) if best else None

    async def live_worker_ids(self) -> list[int]:
        r = await self.conn.rpc(RpcCode.GetFilesystemInfo, {})
        return [w["address"]["worker_id"]
                for w in r.header.get("live_workers", [])]

    async def add_command(self, worker_id: int, cmd: dict) -> None:
        await self.conn.rpc(RpcCode.SubmitTask,
                            {"worker_id": worker_id, "command": cmd})

    async def close(self):
        await self.conn.close()


class _Handler:
    CODES = {int(RpcCode.SubmitJob), int(RpcCode.GetJobStatus),
             int(RpcCode.CancelJob), int(RpcCode.ReportTask),
             int(RpcCode.SubmitTransfer), int(RpcCode.GetTransferStatus),
             int(RpcCode.CancelTransfer), int(RpcCode.ReportTransferTask),
             int(RpcCode.QueryTransferTask), int(RpcCode.ListTransfers),
             int(RpcCode.RetryTransfer), int(RpcCode.Heartbeat)}

    def __init__(self, svc: "TransferService"):
        self.svc = svc

    async def handle(self, msg: Message, conn) -> Optional[Message]:
        code = msg.code
        jm = self.svc.jobs
        h = msg.header
        if code == int(RpcCode.Heartbeat):
            return msg.reply({"service": "transfer"})
        if code in (int(RpcCode.SubmitJob), int(RpcCode.SubmitTransfer)):
            return msg.reply(await self.svc.submit(h))
        if code in (int(RpcCode.GetJobStatus),
                    int(RpcCode.GetTransferStatus)):
            return msg.reply(jm.status(h.get("job_id")
                                       or h.get("transfer_id")))
        if code in (int(RpcCode.CancelJob), int(RpcCode.CancelTransfer)):
            return msg.reply(jm.cancel(h.get("job_id")
                                       or h.get("transfer_id")))
        if code in (int(RpcCode.ReportTask),
                    int(RpcCode.ReportTransferTask)):
            jm.report_task(h)
            return msg.reply({})
        if code == int(RpcCode.QueryTransferTask):
            job = jm.jobs.get(h.get("job_id", ""))
            if job is None:
                raise err.JobNotFound(h.get("job_id", ""))
            return msg.reply({"tasks": list(job["tasks"].values())})
        if code == int(RpcCode.ListTransfers):
            return msg.reply({"jobs": [jm.status(j) for j in jm.jobs]})
        if code == int(RpcCode.RetryTransfer):
            return msg.reply(await self.svc.retry(h.get("job_id")))
        raise err.Unsupported(f"transfer rpc code {code}")


class _Service(HandlerService):
    def __init__(self, svc):
        self.svc = svc

    def get_message_handler(self):
        return _Handler(self.svc)


class TransferService:
    """Own RPC server + store + planner/scheduler; master-agnostic apart
    from the RemoteEnv RPC surface."""

    def __init__(self, conf: ClusterConf, port: int = 0):
        self.conf = conf
        self.env = RemoteEnv(conf)
        self.jobs = JobManager(None, conf=conf)   # env-driven planning
        self.rpc = RpcServer("transfer", conf.master.hostname, port,
                             _Service(self))
        self.port = 0

    async def start(self) -> "TransferService":
        await self.rpc.start()
        self.port = self.rpc.port
        log.info("transfer service on :%d (%d jobs restored)",
                 self.port, len(self.jobs.jobs))
        return self

    async def stop(self) -> None:
        await self.rpc.stop()
        await self.env.close()

    # ---------------- planning over RPC ----------------
    async def submit(self, h: dict) -> dict:
        mount = await self.env.mounts_lookup(h["path"])
        workers = await self.env.live_worker_ids()
        job = self.jobs.new_job(h, mount)
        self.jobs.plan(job, mount, workers)
        for task in job["tasks"].values():
            await self.env.add_command(task["worker"],
                                       {"cmd": "load_task", **task})
        self.jobs.store.save(job)
        return {"job_id": job["job_id"], "state": job["state"],
                "total": job["total"]}

    async def retry(self, job_id: str) -> dict:
        out = self.jobs.retry(job_id)   # master=None: state walk only
        job = self.jobs.jobs[job_id]
        for task in job["tasks"].values():
            if task["state"] == "assigned" and out["retried"]:
                await self.env.add_command(task["worker"],
                                           {"cmd": "load_task", **task})
        return out
