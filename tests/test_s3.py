"""S3 UFS connector against an in-process fake S3 (tests/fake_s3.py):
SigV4-signed requests, ranged reads, ListObjectsV2 pagination, multipart
upload, and cache-mount read-through."""
import asyncio
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from fake_s3 import FakeS3  # noqa: E402


@pytest.fixture
def s3():
    srv = FakeS3(max_keys=3).start()
    yield srv
    srv.stop()


def _props(srv, **extra):
    return {"endpoint": srv.endpoint, "region": "test",
            "access_key": "AK", "secret_key": "SK", **extra}


def test_s3_ufs_roundtrip(s3):
    from curvine_amd.ufs.s3 import S3Ufs

    fs = S3Ufs("s3://bkt/pfx", _props(s3))
    data = os.urandom(100_000)
    w = fs.create("/a/one.bin")
    w.write(data[:40_000])
    w.write(data[40_000:])
    w.close()
    assert s3.objects["bkt/pfx/a/one.bin"] == data

    st = fs.status("a/one.bin")
    assert st["length"] == len(data)
    assert fs.status("a/missing") is None

    r = fs.open("a/one.bin")
    assert r.read(1000) == data[:1000]
    r.seek(99_000)
    assert r.read(5000) == data[99_000:]

    # pagination: 7 objects with max_keys=3 forces 3 list pages
    for i in range(6):
        w = fs.create(f"/list/f{i}")
        w.write(b"x" * (i + 1))
        w.close()
    got = {f["path"]: f["length"] for f in fs.list_files("/list")}
    assert got == {f"/list/f{i}": i + 1 for i in range(6)}

    fs.rename("a/one.bin", "a/two.bin")
    assert fs.status("a/one.bin") is None
    assert fs.status("a/two.bin")["length"] == len(data)
    fs.delete("a/two.bin")
    assert fs.status("a/two.bin") is None


def test_s3_multipart_upload(s3):
    from curvine_amd.ufs.s3 import S3Ufs

    fs = S3Ufs("s3://bkt/mp", _props(s3, multipart_part_size=1 << 20))
    data = os.urandom((3 << 20) + 12345)     # 4 parts (3 full + tail)
    w = fs.create("/big.bin")
    for off in range(0, len(data), 300_000):
        w.write(data[off:off + 300_000])
    assert w.upload_id is not None           # multipart engaged
    w.close()
    assert s3.objects["bkt/mp/big.bin"] == data
    assert not s3.uploads                     # upload session completed

    # small object stays a single PUT
    w = fs.create("/small.bin")
    w.write(b"tiny")
    w.close()
    assert s3.objects["bkt/mp/small.bin"] == b"tiny"


def test_s3_mount_read_through(s3, tmp_path):
    """Mount s3:// into the cache namespace: reads fall through to the
    fake S3 and serve correct bytes."""
    from curvine_amd.testing import MiniCluster
    from curvine_amd.unified import UnifiedFileSystem

    payload = os.urandom(5 << 20)
    s3.objects["bkt/data/model.bin"] = payload

    async def main():
        mc = await MiniCluster(tmp_dir=str(tmp_path / "cv")).start()
        fs = UnifiedFileSystem(mc.client_conf())
        try:
            await fs.mount("/s3", "s3://bkt/data", _props(s3),
                           auto_cache=False)
            st = await fs.file_status("/s3/model.bin")
            assert st.length == len(payload)
            got = await fs.read_all("/s3/model.bin")
            assert got == payload
            names = [f.path for f in await fs.list_status("/s3")]
            assert "/s3/model.bin" in names
            # cache-mode writes stay cache-only (reference CACHE mode)
            await fs.write_all("/s3/cached.bin", b"C" * 1000)
            assert "bkt/data/cached.bin" not in s3.objects
            # fs-mode mount: writes mirror through to S3
            await fs.mount("/s3fs", "s3://bkt/data", _props(s3),
                           cache_mode="fs", auto_cache=False)
            await fs.write_all("/s3fs/out.bin", b"W" * 123456)
            assert s3.objects.get("bkt/data/out.bin") == b"W" * 123456
        finally:
            await fs.close()
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())


def test_s3_load_job_into_cache(s3, tmp_path):
    """BASELINE config[3] flow on the CPU tier: `cv load` pulls S3
    objects into the cache; after the S3 endpoint dies, reads still
    serve from cache."""
    import time

    from curvine_amd.testing import MiniCluster
    from curvine_amd.unified import UnifiedFileSystem

    objs = {f"bkt/warm/f{i}.bin": os.urandom(400_000 + i * 31)
            for i in range(4)}
    s3.objects.update(objs)

    async def main():
        mc = await MiniCluster(tmp_dir=str(tmp_path / "cv")).start()
        fs = UnifiedFileSystem(mc.client_conf())
        try:
            await fs.mount("/warm", "s3://bkt/warm", _props(s3),
                           auto_cache=False)
            job = await fs.submit_job("/warm", recursive=True)
            deadline = time.time() + 30
            while time.time() < deadline:
                st = await fs.job_status(job["job_id"])
                if st["state"] not in ("planning", "running"):
                    break
                await asyncio.sleep(0.2)
            assert st["state"] == "completed", st
            assert st["done"] == len(objs)
            # S3 goes away; the cache must now be authoritative
            s3.stop()
            for key, data in objs.items():
                name = key.rsplit("/", 1)[-1]
                got = await fs.read_all(f"/warm/{name}")
                assert got == data, name
        finally:
            await fs.close()
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())


def test_s3_transient_errors_are_retried(s3):
    """429/5xx and connection failures back off and retry (the S3
    contract for throttling); hard 4xx does not retry."""
    from curvine_amd import errors as err
    from curvine_amd.ufs.s3 import S3Ufs

    ufs = S3Ufs("s3://b/pre", _props(s3))
    ufs._RETRIES = 3
    with ufs.create("obj.bin") as w:
        w.write(b"payload")

    s3.fail_next, s3.fail_status = 2, 503
    assert ufs.open("obj.bin").read(7) == b"payload"   # survives two 503s

    s3.fail_next, s3.fail_status = 2, 429
    assert ufs.open("obj.bin").read(7) == b"payload"

    # more consecutive failures than retries -> typed UfsError
    s3.fail_next, s3.fail_status = 10, 500
    before = s3.requests_seen
    with pytest.raises(err.UfsError):
        ufs.open("obj.bin").read(7)
    assert s3.requests_seen - before == ufs._RETRIES + 1
    s3.fail_next = 0
