"""Azure Blob UFS connector against an in-process fake (SharedKey
signing, List Blobs XML, ranged GET, Put Block/Block List chunked
upload, copy-rename) + cache-mount read-through."""
import asyncio
import os
import sys

import pytest

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))
from fake_azure import ACCOUNT, KEY, FakeAzure  # noqa: E402


@pytest.fixture
def az():
    srv = FakeAzure()
    yield srv
    srv.stop()


def _props(srv):
    return {"account": ACCOUNT, "account_key": KEY,
            "endpoint": f"http://{srv.addr}"}


def test_azure_connector_roundtrip(az):
    from curvine_amd.ufs import get_ufs

    az.blobs["cont/pre/a/one.bin"] = b"1" * 1500
    az.blobs["cont/pre/two.bin"] = b"2" * 900
    fs = get_ufs("az://cont/pre", _props(az))
    files = sorted(f["path"] for f in fs.list_files("/"))
    assert files == ["/a/one.bin", "/two.bin"]
    st = fs.status("/a/one.bin")
    assert st["length"] == 1500 and not st["is_dir"]
    assert fs.status("/a")["is_dir"] is True
    assert fs.status("/nope") is None
    with fs.open("/a/one.bin", offset=1400) as r:
        assert r.read(500) == b"1" * 100
    # chunked upload: > one 8 MiB block
    big = os.urandom(20 << 20)
    w = fs.create("/out/big.bin")
    pos = 0
    while pos < len(big):
        w.write(big[pos:pos + (3 << 20)])
        pos += 3 << 20
    w.close()
    assert az.blobs["cont/pre/out/big.bin"] == big
    # small upload = single Put Blob
    w = fs.create("/out/small.bin")
    w.write(b"tiny")
    w.close()
    assert az.blobs["cont/pre/out/small.bin"] == b"tiny"
    fs.rename("/out/small.bin", "/out/renamed.bin")
    assert "cont/pre/out/renamed.bin" in az.blobs
    assert "cont/pre/out/small.bin" not in az.blobs
    fs.delete("/out", recursive=True)
    assert not any(k.startswith("cont/pre/out/") for k in az.blobs)


def test_azure_mount_read_through(az, tmp_path):
    from curvine_amd.testing import MiniCluster
    from curvine_amd.unified import UnifiedFileSystem

    payload = os.urandom(3 << 20)
    az.blobs["cont/data/model.bin"] = payload

    async def main():
        mc = await MiniCluster(tmp_dir=str(tmp_path / "cv")).start()
        fs = UnifiedFileSystem(mc.client_conf())
        try:
            await fs.mount("/az", "az://cont/data", _props(az),
                           auto_cache=False)
            st = await fs.file_status("/az/model.bin")
            assert st.length == len(payload)
            assert await fs.read_all("/az/model.bin") == payload
            # fs-mode write mirrors through (chunked upload path)
            await fs.mount("/azfs", "az://cont/data", _props(az),
                           cache_mode="fs", auto_cache=False)
            await fs.write_all("/azfs/out.bin", b"W" * 123456)
            assert az.blobs.get("cont/data/out.bin") == b"W" * 123456
        finally:
            await fs.close()
            await mc.stop()

    asyncio.new_event_loop().run_until_complete(main())
