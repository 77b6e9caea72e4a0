"""Lock-order stress (the reference's lock_order_deadlock_stress_test.rs
analog): many threads drive overlapping lock-taking paths — block store
create/finalize/read/delete/demote and shared FUSE writers — under a
deadline; a deadlock or lock-order inversion shows up as a timeout or a
poisoned failure."""
import os
import random
import threading
import time

import pytest

from curvine_amd.conf import WorkerConf
from curvine_amd.worker.block_store import BlockStore


def test_block_store_lock_order_stress(tmp_path):
    conf = WorkerConf(data_dirs=[f"[MEM:128MB]{tmp_path}/mem",
                                 f"[SSD:256MB]{tmp_path}/ssd"])
    store = BlockStore(conf)
    stop = threading.Event()
    errs: list = []
    created = set()
    created_lock = threading.Lock()
    next_id = [1]

    def writer(t):
        rng = random.Random(t)
        try:
            while not stop.is_set():
                with created_lock:
                    bid = next_id[0]
                    next_id[0] += 1
                try:
                    w = store.create_writer(bid, 256 << 10,
                                            rng.choice(["MEM", "SSD"]))
                except Exception:  # noqa: BLE001 — capacity under churn
                    continue
                w.write(os.urandom(64 << 10))
                if rng.random() < 0.1:
                    store.abort(bid)
                else:
                    store.finalize(bid, 64 << 10)
                    with created_lock:
                        created.add(bid)
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    def reader(t):
        rng = random.Random(100 + t)
        try:
            while not stop.is_set():
                with created_lock:
                    if not created:
                        continue
                    bid = rng.choice(list(created))
                try:
                    r = store.open_reader(bid)
                except Exception:  # noqa: BLE001 — deleted meanwhile
                    continue
                r.read(0, 4096)
                r.close()
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    def deleter(t):
        rng = random.Random(200 + t)
        try:
            while not stop.is_set():
                with created_lock:
                    if not created:
                        continue
                    bid = rng.choice(list(created))
                    created.discard(bid)
                store.delete(bid)
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    def demoter(t):
        try:
            while not stop.is_set():
                store.demote_coldest(high_watermark=0.3, low_watermark=0.2)
                store.reap_deferred()
                time.sleep(0.01)
        except Exception as e:  # noqa: BLE001
            errs.append(e)

    threads = ([threading.Thread(target=writer, args=(t,))
                for t in range(4)] +
               [threading.Thread(target=reader, args=(t,))
                for t in range(4)] +
               [threading.Thread(target=deleter, args=(t,))
                for t in range(2)] +
               [threading.Thread(target=demoter, args=(0,))])
    for t in threads:
        t.start()
    time.sleep(4.0)
    stop.set()
    deadline = time.time() + 20
    for t in threads:
        t.join(timeout=max(0.1, deadline - time.time()))
    stuck = [t for t in threads if t.is_alive()]
    assert not stuck, f"{len(stuck)} threads wedged (lock-order deadlock?)"
    assert not errs, errs[0]
    store.close()
