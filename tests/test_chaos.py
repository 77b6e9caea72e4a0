"""Bounded chaos/recovery soak: a paged master + native-data worker
cluster under concurrent mixed clients while a second worker joins,
one worker is lost (heartbeat expiry -> location cleanup ->
re-replication), and TTL/eviction ticks run.  Every surviving file must
read back byte-exact afterwards — the cross-feature integration the
per-feature tests can't see."""
import asyncio
import os
import random

import pytest


def _run(coro):
    loop = asyncio.new_event_loop()
    asyncio.set_event_loop(loop)
    try:
        return loop.run_until_complete(coro)
    finally:
        loop.close()


def test_mixed_chaos_recovery(tmp_path):
    from curvine_amd.testing import MiniCluster
    from curvine_amd.testing import test_conf as tc

    conf = tc(str(tmp_path))
    conf.master.max_resident_inodes = 300       # force paging under load
    conf.master.heartbeat_check_ms = 300
    conf.master.worker_expire_ms = 1500
    conf.worker.heartbeat_interval_ms = 200

    async def main():
        mc = await MiniCluster(conf=conf, tmp_dir=str(tmp_path),
                               workers=2).start()
        fs = mc.fs()
        rng = random.Random(11)
        contents: dict[str, bytes] = {}
        deleted: set[str] = set()

        async def churn(wid: int, n_ops: int):
            for i in range(n_ops):
                op = rng.random()
                p = f"/chaos/w{wid}/f{rng.randrange(40)}"
                try:
                    if op < 0.5:
                        data = os.urandom(rng.randrange(1024, 1 << 20))
                        await fs.write_all(p, data, replicas=2)
                        contents[p] = data
                        deleted.discard(p)
                    elif op < 0.7 and p in contents:
                        got = await fs.read_all(p)
                        assert got == contents[p], f"mid-churn corrupt {p}"
                    elif op < 0.8 and p in contents:
                        await fs.delete(p)
                        del contents[p]
                        deleted.add(p)
                    elif op < 0.9 and p in contents:
                        d = p + ".mv"
                        await fs.rename(p, d)
                        contents[d] = contents.pop(p)
                    else:
                        await fs.mkdir(f"/chaos/d{rng.randrange(50)}",
                                       create_parents=True)
                except Exception as e:  # noqa: BLE001
                    raise AssertionError(f"op on {p} failed: {e}") from e

        # phase 1: concurrent churn across 4 logical clients
        await asyncio.gather(*[churn(w, 60) for w in range(4)])

        # phase 2: kill worker 0 mid-flight churn; replicas must cover
        victim = mc.workers[0]
        churn_task = asyncio.gather(*[churn(10 + w, 30) for w in range(2)])
        await victim.stop()
        await churn_task
        # wait for expiry + location cleanup + (re-)replication
        for _ in range(100):
            await asyncio.sleep(0.1)
            live = mc.master.fs.workers.live_workers()
            if len(live) == 1:
                break
        assert len(mc.master.fs.workers.live_workers()) == 1

        # every surviving file reads back byte-exact from the survivor
        bad = []
        for p, data in sorted(contents.items()):
            try:
                got = await fs.read_all(p)
                if got != data:
                    bad.append((p, "corrupt"))
            except Exception as e:  # noqa: BLE001
                bad.append((p, str(e)))
        assert not bad, f"{len(bad)} of {len(contents)} damaged: {bad[:5]}"
        for p in deleted:
            from curvine_amd import errors as err
            with pytest.raises(err.FsError):
                await fs.read_all(p)

        # phase 3: a fresh worker joins and the cluster keeps mutating
        from curvine_amd.worker.server import Worker
        import copy
        wc = copy.deepcopy(mc.conf)
        wc.worker.rpc_port = 0
        wc.worker.data_dirs = [f"[MEM:64MB]{tmp_path}/w9/mem",
                               f"[SSD:1GB]{tmp_path}/w9/ssd"]
        w9 = await Worker(wc, worker_id=99).start()
        try:
            await churn(20, 30)
            for p, data in list(contents.items())[:10]:
                assert await fs.read_all(p) == data
        finally:
            await w9.stop()
        await fs.close()
        await mc.stop()

    _run(main())
