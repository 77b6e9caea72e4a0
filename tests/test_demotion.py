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


def test_file_tier_capacity_fallthrough(tmp_path):
    """a full SSD dir raises CapacityExceeded from
    FileLayout.allocate so create_writer falls through to the next tier
    instead of overcommitting the disk."""
    from curvine_amd import errors as err

    conf = WorkerConf(data_dirs=[f"[SSD:8MB]{tmp_path}/ssd",
                                 f"[HDD:64MB]{tmp_path}/hdd"])
    store = BlockStore(conf)
    try:
        w1 = store.create_writer(1, 6 << 20, "SSD")
        assert w1.layout.tier == "SSD"
        # SSD has 2 MB left: a 4 MB reservation must land on HDD
        w2 = store.create_writer(2, 4 << 20, "SSD")
        assert w2.layout.tier == "HDD"
        # every dir full -> CapacityExceeded surfaces to the caller
        with pytest.raises(err.CapacityExceeded):
            store.create_writer(3, 256 << 20, "SSD")
        # abort w1: reservation released, SSD takes new writes again
        store.abort(1)
        w4 = store.create_writer(4, 6 << 20, "SSD")
        assert w4.layout.tier == "SSD"
        ssd = next(l for l in store.layouts if l.tier == "SSD")
        # finalize shrinks the reservation to the true length
        w4.write(b"x" * (1 << 20))
        store.finalize(4, 1 << 20)
        assert ssd.used == 1 << 20
    finally:
        store.close()
