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


def test_pread_batch_ptr(tmp_path):
    """Batched fixed-size reads land in the right slots."""
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            data = os.urandom(6 << 20)   # spans two 4MB blocks
            await fs.write_all("/b.bin", data)
            r = await fs.open("/b.bin")
            sr = r.to_sync()
            from curvine_amd.native import PinnedBuffer
            depth = 32
            pbuf = PinnedBuffer(4096 * depth)
            import random
            rng = random.Random(9)
            offs = [rng.randrange(len(data) - 4096) for _ in range(depth)]
            offs[3] = (4 << 20) - 100   # block-spanning -> slow path
            sr.pread_batch_ptr(offs, 4096, pbuf.ptr, 4096)
            for i, off in enumerate(offs):
                got = bytes(pbuf.view[i * 4096:(i + 1) * 4096])
                assert got == data[off:off + 4096], f"slot {i} off {off}"
            pbuf.close()
            sr.close()
            r.close()
            await fs.close()
    run(main())
