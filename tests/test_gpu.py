"""GPU tests (MI355X): device arena, CRC32C kernel vs host, gather kernel,
fill kernel, HBM block store, end-to-end HBM cache path.

Numerics policy: every HIP kernel result is compared against the host
(C++ SSE4.2 / memcpy) implementation of the same op on the same bytes.
"""
import asyncio
import os

import numpy as np
import pytest

from curvine_amd import native

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def dev_arena():
    assert native.gpu_available(), "GPU required: _native must see a device"
    a = native.Arena(0, 256 << 20, staging_bytes=4 << 20, staging_count=4)
    yield a
    a.close()


def test_native_is_device_build():
    assert native.load().__hip_arch__ == "gfx950"
    assert native.device_count() >= 1


def test_arena_roundtrip_large(dev_arena):
    n = 64 << 20
    data = np.random.default_rng(1).integers(0, 256, n, dtype=np.uint8)
    dev_arena.write(0, data, 0, n)
    out = np.zeros(n, dtype=np.uint8)
    dev_arena.read(0, out, 0, n)
    assert (data == out).all()


def test_arena_roundtrip_unaligned_sizes(dev_arena):
    for n in (1, 7, 4095, 4097, 1 << 20, (1 << 20) + 13):
        data = np.random.default_rng(n).integers(0, 256, n, dtype=np.uint8)
        dev_arena.write(4096, data, 0, n)
        out = np.zeros(n, dtype=np.uint8)
        dev_arena.read(4096, out, 0, n)
        assert (data == out).all(), f"size {n}"


def test_crc32c_device_vs_host(dev_arena):
    rng = np.random.default_rng(2)
    for n in (9, 4096, 4100, 65536, (16 << 20) + 123):
        data = rng.integers(0, 256, n, dtype=np.uint8)
        dev_arena.write(0, data, 0, n)
        crc_dev = dev_arena.crc32c(0, n)
        crc_host = native.crc32c(data.tobytes())
        assert crc_dev == crc_host, f"n={n}: {crc_dev:#x} != {crc_host:#x}"


def test_crc32c_known_answer():
    assert native.crc32c(b"123456789") == 0xE3069283


def test_fill_kernel(dev_arena):
    dev_arena.fill(0, 1 << 20, 0xAB)
    out = np.zeros(1 << 20, dtype=np.uint8)
    dev_arena.read(0, out, 0, 1 << 20)
    assert (out == 0xAB).all()
    dev_arena.fill(100, 50, 0)
    dev_arena.read(0, out, 0, 200)
    assert (out[100:150] == 0).all() and (out[99] == 0xAB) and (out[150] == 0xAB)


def test_gather_kernel(dev_arena):
    rng = np.random.default_rng(3)
    data = rng.integers(0, 256, 8 << 20, dtype=np.uint8)
    dev_arena.write(0, data, 0, len(data))
    extents = [(0, 100), (4096, 65536), (1 << 20, 3 << 20), (7 << 20, 999)]
    total = sum(e[1] for e in extents)
    out = np.zeros(total, dtype=np.uint8)
    dev_arena.gather(extents, out, 0)
    expect = np.concatenate([data[o:o + n] for o, n in extents])
    assert (out == expect).all()


def test_device_to_device_copy(dev_arena):
    import torch
    n = 8 << 20
    data = np.random.default_rng(4).integers(0, 256, n, dtype=np.uint8)
    dev_arena.write(0, data, 0, n)
    t = torch.zeros(n, dtype=torch.uint8, device="cuda:0")
    dev_arena.read_to_ptr(0, t.data_ptr(), n, device=True)
    torch.cuda.synchronize()
    assert (t.cpu().numpy() == data).all()
    # and back: device tensor -> arena
    t2 = torch.arange(n, dtype=torch.int32, device="cuda:0").view(torch.uint8)
    dev_arena.write_from_ptr(0, t2.data_ptr(), n * 4, device=True)
    out = np.zeros(n * 4, dtype=np.uint8)
    dev_arena.read(0, out, 0, n * 4)
    assert (out == t2.cpu().numpy()).all()


def test_hbm_block_store():
    from curvine_amd.conf import WorkerConf
    from curvine_amd.worker.block_store import BlockStore
    conf = WorkerConf(data_dirs=["[HBM:256MB:0]gpu0"])
    store = BlockStore(conf)
    try:
        w = store.create_writer(1, 64 << 20, "HBM")
        data = os.urandom(10 << 20)
        w.write(data)
        assert store.finalize(1, len(data)) == "HBM"
        r = store.open_reader(1)
        assert r.read(0, 100) == data[:100]
        assert r.read(5 << 20, 1 << 20) == data[5 << 20:6 << 20]
        assert r.crc32c(0, len(data)) == native.crc32c(data)
        r.close()
        store.delete(1)
        assert store.block_count() == 0
        # capacity is released
        assert store.storages()[0].used == 0
    finally:
        store.close()


def test_e2e_hbm_cache(tmp_path):
    """MiniCluster with an HBM tier: write through client, read back."""
    from curvine_amd.testing import MiniCluster, test_conf

    async def main():
        conf = test_conf(str(tmp_path))
        mc = MiniCluster(conf=conf, tmp_dir=str(tmp_path),
                         worker_dirs=[["[HBM:256MB:0]gpu0",
                                       f"[SSD:1GB]{tmp_path}/ssd"]])
        await mc.start()
        try:
            fs = mc.fs()
            data = os.urandom(24 << 20)
            await fs.write_all("/gpu.bin", data, storage_tier="HBM")
            tiers = {s.tier: s.used for s in mc.workers[0].store.storages()}
            assert tiers.get("HBM", 0) > 0
            back = await fs.read_all("/gpu.bin")
            assert back == data
            await fs.close()
        finally:
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())


def test_lz4_device_decompress(dev_arena):
    """GPU LZ4 decompress into the HBM arena == host decompress."""
    import random
    rng = random.Random(7)
    for i, data in enumerate([
            bytes(rng.choices(b"abcdefgh", k=3_000_000)),
            os.urandom(500_000),
            b"\x00" * (64 << 10) * 5 + os.urandom(1000),
            (b"pattern123" * 7000)[: (64 << 10) * 2 + 13]]):
        comp = native.lz4_compress(data)
        n = native.load().arena_lz4_decompress(dev_arena.handle, 0, comp)
        assert n == len(data)
        out = np.zeros(len(data), dtype=np.uint8)
        dev_arena.read(0, out, 0, len(data))
        assert out.tobytes() == data, f"lz4 case {i} device mismatch"


def test_pread_to_device_mixed_remote(tmp_path):
    """pread_to_device works for remote blocks via pinned staging."""
    import asyncio

    import torch

    from curvine_amd.testing import MiniCluster, test_conf
    from curvine_amd.worker import registry

    async def main():
        conf = test_conf(str(tmp_path))
        mc = MiniCluster(conf=conf, tmp_dir=str(tmp_path),
                         worker_dirs=[["[HBM:128MB:0]gpu0"]])
        await mc.start()
        try:
            fs = mc.fs()
            data = os.urandom(8 << 20)
            await fs.write_all("/mix.bin", data, storage_tier="HBM")
            t = torch.zeros(len(data), dtype=torch.uint8, device="cuda:0")
            r = await fs.open("/mix.bin")
            # force the remote path by hiding the in-process store
            saved = dict(registry._stores)
            registry._stores.clear()
            try:
                n = await r.pread_to_device(0, t.data_ptr(), len(data))
            finally:
                registry._stores.update(saved)
            torch.cuda.synchronize()
            assert n == len(data)
            assert t.cpu().numpy().tobytes() == data
            r.close()
            await fs.close()
        finally:
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())


def test_device_loader_hbm(tmp_path):
    """CurvineDeviceLoader: tar samples cached in the HBM tier gathered
    straight into a cuda tensor (copy_extents_kernel, no host hop)."""
    import io
    import tarfile

    import torch

    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.dataset import CurvineDeviceLoader
    from curvine_amd.testing import SyncMiniCluster, test_conf

    conf = test_conf(str(tmp_path))
    smc = SyncMiniCluster(conf=conf, tmp_dir=str(tmp_path),
                          worker_dirs=[["[HBM:512MB:0]gpu0"]]).start()
    try:
        sf = SyncFs(smc.client_conf())
        samples = {}
        shard_paths = []
        for s in range(2):
            buf = io.BytesIO()
            with tarfile.open(fileobj=buf, mode="w") as tf:
                for i in range(40):
                    name = f"g{s}-{i}"
                    payload = os.urandom(64 << 10)
                    samples[name] = payload
                    info = tarfile.TarInfo(name)
                    info.size = len(payload)
                    tf.addfile(info, io.BytesIO(payload))
            path = f"/gdl/shard-{s}.tar"
            sf.write_file(path, buf.getvalue(), storage_tier="HBM")
            shard_paths.append(path)
        dl = CurvineDeviceLoader(smc.client_conf(), shard_paths,
                                 device="cuda:0", batch_size=16)
        assert dl.num_samples == 80
        seen = {}
        for tensor, sections, names in dl:
            assert tensor.device.type == "cuda"
            host = tensor.cpu().numpy().tobytes()
            for (start, ln), name in zip(sections, names):
                seen[name] = host[start:start + ln]
        assert seen == samples
        dl.close()
        sf.shutdown()
    finally:
        smc.stop()


def test_registered_reader_batch_hbm(tmp_path):
    """Registered-reader batched preads from HBM arenas: bytes must match
    the written data (the IOPS bench measures speed, this checks truth)."""
    import random

    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.client.reader import SyncLocalReader
    from curvine_amd.testing import SyncMiniCluster, test_conf

    conf = test_conf(str(tmp_path))
    smc = SyncMiniCluster(conf=conf, tmp_dir=str(tmp_path),
                          worker_dirs=[["[HBM:512MB:0]gpu0"]]).start()
    try:
        sf = SyncFs(smc.client_conf())
        data = os.urandom(32 << 20)
        sf.write_file("/rr/h.bin", data, storage_tier="HBM")
        fb = sf.call(sf.fs.client.open("/rr/h.bin"))
        r = SyncLocalReader(fb)
        assert r._native_rid is not None
        n = 4096
        rng = random.Random(5)
        offs = [rng.randrange(len(data) - n) for _ in range(256)]
        offs.append(len(data) - n)
        pbuf = native.PinnedBuffer(len(offs) * n)
        assert r.pread_batch_ptr(offs, n, pbuf.ptr, n) == len(offs)
        view = bytes(pbuf.view[:len(offs) * n])
        for i, off in enumerate(offs):
            assert view[i * n:(i + 1) * n] == data[off:off + n], f"@{off}"
        r.close()
        pbuf.close()
        sf.shutdown()
    finally:
        smc.stop()
