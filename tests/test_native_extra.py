"""LZ4 codec host tests + pinned buffers."""
import os
import random

from curvine_amd import native


def test_lz4_roundtrip_host():
    rng = random.Random(3)
    for data in (b"", b"xyz" * 40000, os.urandom(200_000),
                 bytes(rng.choices(b"ab", k=150_000))):
        assert native.lz4_decompress(native.lz4_compress(data)) == data


def test_pinned_buffer():
    b = native.PinnedBuffer(4096)
    b.view[:5] = b"hello"
    assert bytes(b.view[:5]) == b"hello"
    assert b.ptr != 0
    b.close()


def test_registered_reader_batch(tmp_path):
    """Native registered-reader batch preads (MEM tier, host arenas):
    results match per-read preads; spanning/EOF reads fall back."""
    import os
    import random

    from curvine_amd import native
    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.client.reader import SyncLocalReader
    from curvine_amd.testing import SyncMiniCluster, test_conf

    conf = test_conf(str(tmp_path))
    conf.master.block_size = 1 << 20
    conf.client.block_size = 1 << 20
    smc = SyncMiniCluster(conf=conf, tmp_dir=str(tmp_path)).start()
    try:
        sf = SyncFs(smc.client_conf())
        data = os.urandom((3 << 20) + 777)
        sf.write_file("/rb/f", data)
        fb = sf.call(sf.fs.client.open("/rb/f"))
        r = SyncLocalReader(fb)
        assert r._native_rid is not None, "registration should engage"
        n = 4096
        rng = random.Random(11)
        offs = [rng.randrange(len(data) - n) for _ in range(128)]
        offs += [(1 << 20) - 100]          # spans block boundary: fallback
        offs += [len(data) - n]            # clamped tail
        buf = native.PinnedBuffer(len(offs) * n)
        assert r.pread_batch_ptr(offs, n, buf.ptr, n) == len(offs)
        view = bytes(buf.view[:len(offs) * n])
        for i, off in enumerate(offs):
            assert view[i * n:(i + 1) * n] == data[off:off + n], f"@{off}"
        r.close()
        buf.close()
        sf.shutdown()
    finally:
        smc.stop()
