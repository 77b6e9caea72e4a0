"""Worker decommission drain + mutation retry-cache dedup."""
import asyncio
import os

import pytest

from curvine_amd.rpc.codes import RpcCode
from curvine_amd.testing import MiniCluster


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_decommission_drains_blocks(tmp_path):
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path), workers=2) as mc:
            fs = mc.fs()
            data = os.urandom(2 << 20)
            await fs.write_all("/drain.bin", data, replicas=1)
            await asyncio.sleep(0.3)
            fb = await fs.client.open("/drain.bin")
            holder = fb.blocks[0].locations[0].worker_id
            r = await fs.client.connector.rpc(RpcCode.DecommissionWorker,
                                              {"worker_id": holder})
            assert r.header["state"] == "decommissioning"
            # replication manager must copy the block to the other worker
            other = next(w.worker_id for w in mc.workers
                         if w.worker_id != holder)
            for _ in range(100):
                await asyncio.sleep(0.2)
                mc.master.replication.check_all()
                mc.master.replication.scan()
                fb = await fs.client.open("/drain.bin")
                wids = {a.worker_id for a in fb.blocks[0].locations}
                if other in wids:
                    break
            assert other in wids, f"block not drained: {wids}"
            assert await fs.read_all("/drain.bin") == data
            await fs.close()
    run(main())


def test_retry_cache_dedups_add_block(tmp_path):
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            await fs.client.create("/dup.bin")
            conn = fs.client.connector
            header = {"path": "/dup.bin", "commit_prev_len": -1,
                      "client_host": "", "client_worker_id": -1,
                      "exclude_workers": [], "cid": "test-cid", "rid": 77}
            r1 = await conn.rpc(RpcCode.AddBlock, dict(header))
            # identical (cid, rid): replayed request, must NOT allocate again
            r2 = await conn.rpc(RpcCode.AddBlock, dict(header))
            assert r1.header["block"]["block"]["block_id"] == \
                r2.header["block"]["block"]["block_id"]
            # new rid -> new block
            header["rid"] = 78
            r3 = await conn.rpc(RpcCode.AddBlock, dict(header))
            assert r3.header["block"]["block"]["block_id"] != \
                r1.header["block"]["block"]["block_id"]
            await fs.close()
    run(main())
