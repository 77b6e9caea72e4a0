"""End-to-end CRC32C integrity: workers stamp a publish-time block CRC,
enable_crc clients cross-check their running CRC at commit, and readers
can verify resident blocks."""
import asyncio
import os

import pytest

from curvine_amd import errors as err
from curvine_amd.testing import MiniCluster
from curvine_amd.testing import test_conf as _test_conf


def _run(coro):
    return asyncio.new_event_loop().run_until_complete(coro)


def test_write_crc_crosscheck_and_verify(tmp_path):
    async def main():
        conf = _test_conf(str(tmp_path))
        conf.master.block_size = 1 << 20
        conf.client.block_size = 1 << 20
        conf.client.enable_crc = True
        mc = await MiniCluster(conf=conf, tmp_dir=str(tmp_path)).start()
        fs = mc.fs()
        data = os.urandom((2 << 20) + 999)
        await fs.write_all("/crc/ok", data)      # multi-block, crc checked
        assert await fs.read_all("/crc/ok") == data

        # reader-side verify: clean blocks pass
        from curvine_amd.client.reader import SyncLocalReader
        fb = await fs.client.open("/crc/ok")
        r = SyncLocalReader(fb)
        assert r.verify() == []

        # corrupt a block in the store behind the reader's back
        store = mc.workers[0].store
        bid = fb.blocks[0].block.block_id
        w = store.reopen_writer(bid)
        # reopen drops the stored crc; re-stamp the OLD one to simulate
        # silent corruption rather than an audited rewrite
        old_crc = None
        w.pwrite(10, b"\x00\x00\x00\x00CORRUPT")
        with store.lock:
            store.blocks[bid].meta["crc32c"] = (
                r._readers[0].crc32c(0, fb.blocks[0].block.length) ^ 0xDEAD)
        assert r.verify() == [bid]
        r.close()
        await fs.close()
        await mc.stop()

    _run(main())


def test_commit_crc_mismatch_detected(tmp_path):
    """Bytes mutated between write and commit: the client's running CRC
    disagrees with the worker's publish-time CRC -> ChecksumMismatch."""
    async def main():
        conf = _test_conf(str(tmp_path))
        conf.client.enable_crc = True
        mc = await MiniCluster(conf=conf, tmp_dir=str(tmp_path)).start()
        fs = mc.fs()
        w = await fs.create("/crc/bad", overwrite=True)
        await w.write(b"A" * 100_000)
        await w.flush()   # push the buffer so the block exists
        # corrupt the in-flight block before complete() publishes it
        bid = w._block.block.block_id
        store = mc.workers[0].store
        store.reopen_writer(bid).pwrite(0, b"ZZZZ")
        with pytest.raises(err.ChecksumMismatch):
            await w.complete()
        await fs.close()
        await mc.stop()

    _run(main())
