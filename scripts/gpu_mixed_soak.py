#!/usr/bin/env python3
"""Mixed-workload endurance soak: concurrent writers, verifying readers,
deleters and metadata scanners against one HBM-tier cluster for N
seconds.  Every read is CRC-checked against the writer's record; any
mismatch or unexpected error fails the run.

Usage: python scripts/gpu_mixed_soak.py [--seconds 300] [--file-mb 128]
"""
import argparse
import asyncio
import json
import os
import random
import sys
import time
import zlib

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from curvine_amd import errors as cverr  # noqa: E402


async def main(args):
    import tempfile

    from curvine_amd.testing import MiniCluster, test_conf
    from curvine_amd import native

    tier = "HBM" if native.gpu_available() else "MEM"
    tmp = tempfile.mkdtemp(prefix="mixed-soak-")
    conf = test_conf(tmp)
    cap = args.capacity_gb
    conf.worker.data_dirs = (
        [f"[HBM:{cap}GB:0]gpu0"] if tier == "HBM"
        else [f"[MEM:{min(cap, 2)}GB]{tmp}/mem"])
    stats = {"writes": 0, "reads": 0, "deletes": 0, "stats": 0,
             "read_bytes": 0, "write_bytes": 0, "tolerated_races": 0}
    published: dict[str, int] = {}     # path -> crc32 (stable files)
    stop = time.monotonic() + args.seconds
    errors: list[str] = []

    async with MiniCluster(conf=conf, tmp_dir=tmp) as mc:
        fs = mc.fs()
        base = os.urandom(args.file_mb << 20)
        base_crc = zlib.crc32(base)

        async def writer(t):
            i = 0
            while time.monotonic() < stop and not errors:
                if len(published) > args.max_files:
                    await asyncio.sleep(0.05)
                    continue
                path = f"/soak/w{t}_{i}"
                i += 1
                await fs.write_all(path, base, storage_tier=tier)
                published[path] = base_crc
                stats["writes"] += 1
                stats["write_bytes"] += len(base)

        async def reader(t):
            while time.monotonic() < stop and not errors:
                if not published:
                    await asyncio.sleep(0.02)
                    continue
                path = random.choice(list(published))
                want = published.get(path)
                try:
                    data = await fs.read_all(path)
                except (cverr.FileNotFound, cverr.BlockNotFound):
                    stats["tolerated_races"] += 1   # deleted under us
                    continue
                if want is None:
                    stats["tolerated_races"] += 1
                    continue
                if zlib.crc32(data) != want:
                    errors.append(f"CRC mismatch on {path}")
                    return
                stats["reads"] += 1
                stats["read_bytes"] += len(data)

        async def deleter():
            while time.monotonic() < stop and not errors:
                if len(published) < args.max_files // 2:
                    await asyncio.sleep(0.05)
                    continue
                path = random.choice(list(published))
                published.pop(path, None)
                try:
                    await fs.delete(path)
                except cverr.FileNotFound:
                    pass
                stats["deletes"] += 1

        async def statter():
            while time.monotonic() < stop and not errors:
                try:
                    await fs.list_status("/soak")
                except cverr.FileNotFound:
                    pass
                for p in random.sample(list(published),
                                       min(8, len(published))):
                    try:
                        await fs.file_status(p)
                    except cverr.FileNotFound:
                        pass
                stats["stats"] += 1
                await asyncio.sleep(0.01)

        await fs.mkdir("/soak")
        tasks = ([writer(t) for t in range(args.writers)] +
                 [reader(t) for t in range(args.readers)] +
                 [deleter(), statter()])
        await asyncio.gather(*tasks)
        await fs.close()

    out = {"seconds": args.seconds, "tier": tier, "file_mb": args.file_mb,
           "ok": not errors, "errors": errors, **stats,
           "read_GiBps": round(stats["read_bytes"] / args.seconds / 2**30, 2),
           "write_GiBps": round(stats["write_bytes"] / args.seconds / 2**30,
                                2)}
    print(json.dumps(out))
    return 0 if not errors else 1


if __name__ == "__main__":
    p = argparse.ArgumentParser()
    p.add_argument("--seconds", type=int, default=300)
    p.add_argument("--file-mb", type=int, default=128)
    p.add_argument("--capacity-gb", type=int, default=24)
    p.add_argument("--max-files", type=int, default=100)
    p.add_argument("--writers", type=int, default=3)
    p.add_argument("--readers", type=int, default=4)
    args = p.parse_args()
    sys.exit(asyncio.new_event_loop().run_until_complete(main(args)))
