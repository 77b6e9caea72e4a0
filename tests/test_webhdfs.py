"""WebHDFS UFS connector (curvine_amd/ufs/webhdfs.py) against an
in-process fake WebHDFS server: direct connector ops, mount
read-through, write mirror, and the load-job ingest + cached-read flow
(the reference covers hdfs:// via OpenDAL+JVM; see webhdfs.py)."""
import asyncio
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from fake_webhdfs import FakeWebHdfs  # noqa: E402


@pytest.fixture
def hdfs():
    srv = FakeWebHdfs()
    yield srv
    srv.stop()


def _uri(srv, base="/data"):
    return f"hdfs://{srv.addr}{base}"


def test_connector_roundtrip(hdfs):
    from curvine_amd.ufs import get_ufs

    hdfs.put("/data/a/one.bin", b"1" * 1000)
    hdfs.put("/data/a/two.bin", b"2" * 2000)
    fs = get_ufs(_uri(hdfs), {"user": "alice"})
    files = sorted(f["path"] for f in fs.list_files("/"))
    assert files == ["/a/one.bin", "/a/two.bin"]
    st = fs.status("/a/two.bin")
    assert st["length"] == 2000 and not st["is_dir"]
    assert fs.status("/nope") is None
    with fs.open("/a/one.bin") as r:
        assert r.read(100) == b"1" * 100
    with fs.open("/a/two.bin", offset=1990) as r:
        assert r.read(100) == b"2" * 10
    # write (CREATE chunk + APPEND chunks) with > one chunk of data
    big = os.urandom(20 << 20)
    w = fs.create("/out/bin.dat")
    pos = 0
    while pos < len(big):
        w.write(big[pos:pos + (3 << 20)])
        pos += 3 << 20
    w.close()
    assert hdfs.files["/data/out/bin.dat"] == big
    # the redirect dance really happened
    assert any("PUT CREATE" in r and "dn=True" in r for r in hdfs.requests)
    assert any("POST APPEND" in r and "dn=True" in r for r in hdfs.requests)
    fs.rename("/out/bin.dat", "/out/renamed.dat")
    assert "/data/out/renamed.dat" in hdfs.files
    fs.delete("/out/renamed.dat")
    assert "/data/out/renamed.dat" not in hdfs.files


def test_hdfs_mount_read_through(hdfs, tmp_path):
    from curvine_amd.testing import MiniCluster
    from curvine_amd.unified import UnifiedFileSystem

    payload = os.urandom(3 << 20)
    hdfs.put("/data/model.bin", payload)

    async def main():
        mc = await MiniCluster(tmp_dir=str(tmp_path / "cv")).start()
        fs = UnifiedFileSystem(mc.client_conf())
        try:
            await fs.mount("/h", _uri(hdfs), {"user": "alice"},
                           auto_cache=False)
            st = await fs.file_status("/h/model.bin")
            assert st.length == len(payload)
            assert await fs.read_all("/h/model.bin") == payload
            names = [f.path for f in await fs.list_status("/h")]
            assert "/h/model.bin" in names
            # fs-mode mount mirrors writes through to HDFS
            await fs.mount("/hfs", _uri(hdfs), {"user": "alice"},
                           cache_mode="fs", auto_cache=False)
            await fs.write_all("/hfs/out.bin", b"W" * 123456)
            assert hdfs.files.get("/data/out.bin") == b"W" * 123456
        finally:
            await fs.close()
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())


def test_hdfs_load_job_into_cache(hdfs, tmp_path):
    """cv load against hdfs://: ingest into the cache, then reads keep
    serving after the namenode dies."""
    import time

    from curvine_amd.testing import MiniCluster
    from curvine_amd.unified import UnifiedFileSystem

    objs = {f"/data/warm/f{i}.bin": os.urandom(300_000 + i * 17)
            for i in range(3)}
    for p, b in objs.items():
        hdfs.put(p, b)

    async def main():
        mc = await MiniCluster(tmp_dir=str(tmp_path / "cv")).start()
        fs = UnifiedFileSystem(mc.client_conf())
        try:
            await fs.mount("/warm", _uri(hdfs, "/data/warm"),
                           {"user": "alice"}, auto_cache=False)
            job = await fs.submit_job("/warm", recursive=True)
            deadline = time.time() + 30
            st = {}
            while time.time() < deadline:
                st = await fs.job_status(job["job_id"])
                if st["state"] not in ("planning", "running"):
                    break
                await asyncio.sleep(0.2)
            assert st.get("state") == "completed", st
            hdfs.stop()   # namenode gone: cache must serve
            for p, b in objs.items():
                name = p.rsplit("/", 1)[1]
                got = await fs.read_all(f"/warm/{name}")
                assert got == b, name
        finally:
            await fs.close()
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())
