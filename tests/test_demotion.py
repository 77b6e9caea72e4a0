"""Tier demotion: cold blocks move MEM -> SSD under pressure; data stays
readable; heartbeat reports the new tier."""
import asyncio
import os

import pytest

from curvine_amd.conf import WorkerConf
from curvine_amd.worker.block_store import BlockStore


def test_demote_coldest(tmp_path):
    conf = WorkerConf(data_dirs=[f"[MEM:16MB]{tmp_path}/mem",
                                 f"[SSD:1GB]{tmp_path}/ssd"])
    store = BlockStore(conf)
    try:
        payloads = {}
        for bid in (1, 2, 3):
            data = os.urandom(4 << 20)
            payloads[bid] = data
            w = store.create_writer(bid, 4 << 20, "MEM")
            w.write(data)
            store.finalize(bid, len(data))
        # touch block 3 so 1 and 2 are the cold ones
        r = store.open_reader(3)
        r.read(0, 10)
        r.close()
        mem = next(l for l in store.layouts if l.tier == "MEM")
        assert mem.used >= 12 << 20
        moved = store.demote_coldest(high_watermark=0.6, low_watermark=0.3)
        assert moved >= 2
        tiers = {b["block_id"]: b["tier"] for b in store.full_report()}
        assert tiers[3] == "MEM"            # hot block stayed
        assert tiers[1] == "SSD" and tiers[2] == "SSD"
        for bid, data in payloads.items():
            r = store.open_reader(bid)
            assert r.read(0, len(data)) == data
            r.close()
        # deltas include the tier change
        added, _ = store.take_deltas()
        assert any(a["block_id"] in (1, 2) and a["tier"] == "SSD"
                   for a in added)
    finally:
        store.close()


def test_demote_skips_active_readers(tmp_path):
    conf = WorkerConf(data_dirs=[f"[MEM:8MB]{tmp_path}/mem",
                                 f"[SSD:1GB]{tmp_path}/ssd"])
    store = BlockStore(conf)
    try:
        data = os.urandom(4 << 20)
        w = store.create_writer(1, 4 << 20, "MEM")
        w.write(data)
        store.finalize(1, len(data))
        r = store.open_reader(1)   # held open
        moved = store.demote_coldest(high_watermark=0.4, low_watermark=0.1)
        assert moved == 0
        assert r.read(100, 50) == data[100:150]
        r.close()
        moved = store.demote_coldest(high_watermark=0.4, low_watermark=0.1)
        assert moved == 1
    finally:
        store.close()
