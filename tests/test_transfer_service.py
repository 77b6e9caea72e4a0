"""Standalone data-transfer service (curvine_amd/transfer/service.py):
own process shape (own RPC server + sqlite store), master proxying the
job surface, task dispatch through master heartbeat commands, retry, and
store-backed restart (curvine-data-transfer analog)."""
import asyncio
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from fake_webhdfs import FakeWebHdfs  # noqa: E402


def _run(coro):
    loop = asyncio.new_event_loop()
    asyncio.set_event_loop(loop)
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_standalone_transfer_service(tmp_path):
    from curvine_amd.testing import MiniCluster
    from curvine_amd.transfer import TransferService
    from curvine_amd.unified import UnifiedFileSystem

    hdfs = FakeWebHdfs()
    objs = {f"/data/w/f{i}.bin": os.urandom(200_000 + i * 13)
            for i in range(4)}
    for p, b in objs.items():
        hdfs.put(p, b)

    async def main():
        mc = await MiniCluster(tmp_dir=str(tmp_path / "cv")).start()
        # standalone service with a sqlite store, pointed at the master
        import copy
        sconf = copy.deepcopy(mc.client_conf())
        sconf.job.store = "sqlite"
        sconf.job.store_path = str(tmp_path / "transfer.db")
        svc = await TransferService(sconf).start()
        # master proxies the job surface to the service
        mc.master.conf.job.service_addr = \
            f"127.0.0.1:{svc.port}"
        fs = UnifiedFileSystem(mc.client_conf())
        try:
            await fs.mount("/w", f"hdfs://{hdfs.addr}/data/w", {},
                           auto_cache=False)
            job = await fs.submit_job("/w", recursive=True)
            assert job["total"] == 4
            import time
            deadline = time.time() + 30
            st = {}
            while time.time() < deadline:
                st = await fs.job_status(job["job_id"])
                if st["state"] not in ("planning", "running"):
                    break
                await asyncio.sleep(0.2)
            assert st.get("state") == "completed", st
            # the SERVICE owns the state: ask it directly too
            from curvine_amd.rpc.client import RpcClient
            from curvine_amd.rpc.codes import RpcCode
            c = await RpcClient("127.0.0.1", svc.port,
                                timeout_ms=5000).connect()
            r = await c.rpc(RpcCode.GetTransferStatus,
                            {"job_id": job["job_id"]})
            assert r.header["state"] == "completed"
            r = await c.rpc(RpcCode.ListTransfers, {})
            assert any(j["job_id"] == job["job_id"]
                       for j in r.header["jobs"])
            r = await c.rpc(RpcCode.QueryTransferTask,
                            {"job_id": job["job_id"]})
            assert len(r.header["tasks"]) == 4
            await c.close()
            # cached data serves after the namenode dies
            hdfs.stop()
            for p, b in objs.items():
                name = p.rsplit("/", 1)[1]
                assert await fs.read_all(f"/w/{name}") == b
        finally:
            await fs.close()
            await svc.stop()
            await mc.stop()
        # restart the service: jobs restored from the sqlite store
        svc2 = await TransferService(sconf).start()
        try:
            assert job["job_id"] in svc2.jobs.jobs
            assert svc2.jobs.status(job["job_id"])["state"] == "completed"
        finally:
            await svc2.stop()

    _run(main())
