"""Small-file batch write path (CreateFilesBatch/AddBlocksBatch/
WriteBlocksBatch/CompleteFilesBatch)."""
import asyncio
import os

import pytest

from curvine_amd.testing import MiniCluster


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_batch_write_local_short_circuit(tmp_path):
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            fs.client.local_worker_id = mc.workers[0].worker_id
            files = {f"/batch/f{i:03d}": os.urandom(1000 + i)
                     for i in range(50)}
            files["/batch/empty"] = b""
            sts = await fs.write_files_batch(files)
            assert len(sts) == 51
            for p, data in files.items():
                assert await fs.read_all(p) == data
            await fs.close()
    run(main())


def test_batch_write_remote(tmp_path):
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            conf = mc.client_conf()
            conf.client.short_circuit = False
            from curvine_amd.client.filesystem import CurvineFileSystem
            from curvine_amd.worker import registry
            fs = CurvineFileSystem(conf)
            # hide the in-process store so the RPC batch path runs
            saved = dict(registry._stores)
            registry._stores.clear()
            try:
                files = {f"/rb/f{i}": os.urandom(5000) for i in range(20)}
                await fs.write_files_batch(files)
            finally:
                registry._stores.update(saved)
            for p, data in files.items():
                assert await fs.read_all(p) == data
            await fs.close()
    run(main())
