"""Concurrency stress: block-store state transitions under thread barriers,
reservation rollback races, p95 latency assertion, parallel client IO.

Analog of the reference's discipline (block_store.rs:341-704
Barrier tests, lock_order_deadlock_stress_test.rs, resize_lock_p95_test.rs).
"""
import asyncio
import os
import threading
import time

import pytest

from curvine_amd import errors as err
from curvine_amd.conf import WorkerConf
from curvine_amd.worker.block_store import BlockStore


@pytest.fixture
def store(tmp_path):
    s = BlockStore(WorkerConf(data_dirs=[f"[MEM:256MB]{tmp_path}/mem"]))
    yield s
    s.close()


def test_concurrent_create_same_block(store):
    """Exactly one of N racing writers wins the reservation."""
    n = 16
    barrier = threading.Barrier(n)
    wins, losses = [], []

    def racer(i):
        barrier.wait()
        try:
            store.create_writer(99, 1 << 20, "MEM")
            wins.append(i)
        except err.FsError:
            losses.append(i)

    ths = [threading.Thread(target=racer, args=(i,)) for i in range(n)]
    for t in ths:
        t.start()
    for t in ths:
        t.join()
    assert len(wins) == 1 and len(losses) == n - 1


def test_concurrent_readers_with_delete(store):
    """Delete during active reads defers until the last reader closes."""
    data = os.urandom(1 << 20)
    w = store.create_writer(5, 1 << 20, "MEM")
    w.write(data)
    store.finalize(5, len(data))
    n = 8
    barrier = threading.Barrier(n + 1)
    errors = []

    def reader():
        try:
            r = store.open_reader(5)
            barrier.wait()
            for _ in range(50):
                off = int.from_bytes(os.urandom(2), "little") % (len(data) - 64)
                assert r.read(off, 64) == data[off:off + 64]
            r.close()
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    ths = [threading.Thread(target=reader) for _ in range(n)]
    for t in ths:
        t.start()
    barrier.wait()
    store.delete(5)   # readers still active -> deferred
    for t in ths:
        t.join()
    assert not errors
    assert store.block_count() == 0   # delete applied after last close
    mem = store.layouts[0]
    assert mem.used == 0


def test_abort_rollback_releases_capacity(store):
    mem = store.layouts[0]
    before = mem.used
    n = 12
    barrier = threading.Barrier(n)

    def writer(i):
        barrier.wait()
        w = store.create_writer(1000 + i, 4 << 20, "MEM")
        w.write(b"x" * 1000)
        store.abort(1000 + i)

    ths = [threading.Thread(target=writer, args=(i,)) for i in range(n)]
    for t in ths:
        t.start()
    for t in ths:
        t.join()
    assert mem.used == before
    assert store.block_count() == 0


def test_write_read_parallel_blocks(store):
    """N threads each own a block: no cross-talk, all bytes survive."""
    n = 10
    payloads = {i: os.urandom(512 << 10) for i in range(n)}
    errors = []

    def worker(i):
        try:
            w = store.create_writer(i + 1, 1 << 20, "MEM")
            for off in range(0, len(payloads[i]), 64 << 10):
                w.write(payloads[i][off:off + (64 << 10)])
            store.finalize(i + 1, len(payloads[i]))
            r = store.open_reader(i + 1)
            assert r.read(0, len(payloads[i])) == payloads[i]
            r.close()
        except Exception as e:  # noqa: BLE001
            errors.append(e)

    ths = [threading.Thread(target=worker, args=(i,)) for i in range(n)]
    for t in ths:
        t.start()
    for t in ths:
        t.join()
    assert not errors


def test_master_resize_latency_p95(tmp_path):
    """resize (truncate) stays fast under concurrent open load
    (resize_lock_p95_test.rs analog, generous CI bound)."""
    from curvine_amd.testing import MiniCluster

    async def main():
        async with MiniCluster(tmp_dir=str(tmp_path)) as mc:
            fs = mc.fs()
            await fs.write_all("/p95.bin", os.urandom(4 << 20))
            for i in range(20):
                await fs.write_all(f"/load/f{i}", b"x" * 1000)

            stop = []

            async def churn():
                i = 0
                while not stop:
                    await fs.client.open(f"/load/f{i % 20}")
                    i += 1
            task = asyncio.get_event_loop().create_task(churn())
            lats = []
            for i in range(60):
                t0 = time.perf_counter()
                await fs.resize("/p95.bin", (4 << 20) - i)
                lats.append(time.perf_counter() - t0)
            stop.append(1)
            await asyncio.sleep(0)
            task.cancel()
            lats.sort()
            p95 = lats[int(len(lats) * 0.95)]
            assert p95 < 0.5, f"resize p95 {p95 * 1000:.1f}ms"
            await fs.close()
    asyncio.new_event_loop().run_until_complete(main())
