"""Unified FS: mount routing, UFS fallthrough, auto-cache, fallback reads,
write-through cache."""
import asyncio
import os

import pytest

from curvine_amd import errors as err
from curvine_amd.testing import MiniCluster
from curvine_amd.unified import UnifiedFileSystem


@pytest.fixture
def loop():
    loop = asyncio.new_event_loop()
    asyncio.set_event_loop(loop)
    yield loop
    loop.close()


def run(loop, coro):
    return loop.run_until_complete(coro)


@pytest.fixture
def env(tmp_path, loop):
    ufs_root = tmp_path / "ufs"
    (ufs_root / "d").mkdir(parents=True)
    files = {}
    for i in range(3):
        data = os.urandom(200_000 + i * 17)
        (ufs_root / "d" / f"f{i}.bin").write_bytes(data)
        files[f"f{i}.bin"] = data
    mc = MiniCluster(tmp_dir=str(tmp_path / "cv"))
    run(loop, mc.start())
    conf = mc.client_conf()
    fs = UnifiedFileSystem(conf)
    run(loop, fs.mount("/mnt", f"file://{ufs_root}"))
    yield fs, mc, ufs_root, files
    run(loop, fs.close())
    run(loop, mc.stop())


def test_ufs_fallthrough_read(loop, env):
    fs, mc, ufs_root, files = env

    async def main():
        # not cached: read falls through to UFS
        data = await fs.read_all("/mnt/d/f0.bin")
        assert data == files["f0.bin"]
        st = await fs.file_status("/mnt/d/f1.bin")
        assert st.length == len(files["f1.bin"])
        assert await fs.exists("/mnt/d/f2.bin")
        assert not await fs.exists("/mnt/d/nope.bin")
    run(loop, main())


def test_list_merges_cache_and_ufs(loop, env):
    fs, mc, ufs_root, files = env

    async def main():
        await fs.mkdir("/mnt/d", create_parents=True)
        await fs.write_all("/mnt/d/cached.bin", b"incv")
        names = [s.name for s in await fs.list_status("/mnt/d")]
        assert "cached.bin" in names
        for f in files:
            assert f in names
    run(loop, main())


def test_auto_cache_on_miss(loop, env):
    fs, mc, ufs_root, files = env

    async def main():
        data = await fs.read_all("/mnt/d/f0.bin")
        assert data == files["f0.bin"]
        # auto-cache job should load the file into cv
        for _ in range(80):
            await asyncio.sleep(0.1)
            try:
                fb = await fs.client.open("/mnt/d/f0.bin")
                if fb.status.is_complete and fb.blocks and \
                        all(b.locations for b in fb.blocks):
                    break
            except err.FileNotFound:
                continue
        fb = await fs.client.open("/mnt/d/f0.bin")
        assert fb.status.length == len(files["f0.bin"])
        # second read now comes from cache (still correct)
        assert await fs.read_all("/mnt/d/f0.bin") == files["f0.bin"]
    run(loop, main())


def test_write_through_to_ufs(loop, tmp_path):
    ufs_root = tmp_path / "ufs2"
    ufs_root.mkdir()

    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path / "cv2")) as mc:
            fs = UnifiedFileSystem(mc.client_conf())
            await fs.mount("/m", f"file://{ufs_root}", cache_mode="fs")
            payload = os.urandom(123_456)
            w = await fs.create("/m/out/data.bin", overwrite=True)
            await w.write(payload)
            await w.complete()
            # landed in the UFS too
            assert (ufs_root / "out" / "data.bin").read_bytes() == payload
            # and is readable from cache
            assert await fs.read_all("/m/out/data.bin") == payload
            await fs.delete("/m/out/data.bin")
            assert not (ufs_root / "out" / "data.bin").exists()
            await fs.close()
    loop = asyncio.new_event_loop()
    loop.run_until_complete(main())
    loop.close()


def test_fallback_mid_read(loop, env):
    fs, mc, ufs_root, files = env

    async def main():
        # cache the file, then kill the worker: reads fall back to UFS
        await fs.read_all("/mnt/d/f1.bin")
        for _ in range(80):
            await asyncio.sleep(0.1)
            try:
                fb = await fs.client.open("/mnt/d/f1.bin")
                if fb.status.is_complete and fb.blocks and \
                        all(b.locations for b in fb.blocks):
                    break
            except err.FileNotFound:
                pass
        w = mc.workers[0]
        await w.stop()
        mc.workers.remove(w)
        data = await fs.read_all("/mnt/d/f1.bin")
        assert data == files["f1.bin"]
    run(loop, main())
