#!/usr/bin/env python3
"""Beyond-RAM namespace demo/benchmark (RocksInodeStore-paging analog).

Builds N files with a bounded resident inode map, then measures:
  * RSS and resident-map size vs the same namespace unpaged,
  * stat QPS over the RPC path for hot (resident) and cold (faulted)
    lookups.

Usage: python scripts/paged_meta_bench.py [--n 500000] [--resident 50000]
"""
import argparse
import asyncio
import json
import os
import random
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def rss_mb() -> float:
    with open("/proc/self/status") as f:
        for line in f:
            if line.startswith("VmRSS"):
                return int(line.split()[1]) / 1024.0
    return 0.0


async def build(master, n, out):
    t0 = time.perf_counter()
    fs = master.fs
    for d in range(n // 1000):
        fs.mkdir(f"/pg/d{d}", create_parents=True)
    for i in range(n):
        fs.create(f"/pg/d{i // 1000}/f{i}", 0, 1, "", False)
        fs.complete_file(f"/pg/d{i // 1000}/f{i}", i % 4096, [i % 4096])
        if i % 200_000 == 0 and master.inode_db is not None:
            # interleave flush+evict the way the actor tick does
            while master.inode_db.flush(master.fs.fs_dir,
                                        master.mounts.to_snapshot(),
                                        master.journal.op_id):
                pass
            maxres = master.conf.master.max_resident_inodes
            if maxres:
                master.inode_db.page_out(master.fs.fs_dir,
                                         set(fs.writing), maxres)
    while master.inode_db is not None and (
            master.inode_db.flush(master.fs.fs_dir,
                                  master.mounts.to_snapshot(),
                                  master.journal.op_id)
            or master.inode_db._dirty):
        pass
    maxres = master.conf.master.max_resident_inodes
    if maxres and master.inode_db is not None:
        on_evict = None
        if master.native_meta is not None:
            nm = master.native_meta
            on_evict = lambda iid: nm.lib.meta_drop(nm.sid, iid)
        master.inode_db.page_out(master.fs.fs_dir, set(fs.writing),
                                 maxres, on_evict)
    out["build_s"] = round(time.perf_counter() - t0, 1)
    out["rss_mb_after_build"] = round(rss_mb(), 1)
    out["resident_inodes"] = len(master.fs.fs_dir.inodes)


async def stat_qps(port, n, k, seed, label, out, lo=0):
    """Drive stats from SUBPROCESS clients so the measurement is server
    capacity, not shared-GIL contention with the master."""
    import subprocess
    procs = []
    np = 4
    for p in range(np):
        procs.append(subprocess.Popen(
            [sys.executable, os.path.abspath(__file__), "--stat-worker",
             "--master", f"127.0.0.1:{port}", "--n", str(n),
             "--lo", str(lo),
             "--k", str(k // np), "--seed", str(seed * 1000 + p)],
            stdout=subprocess.PIPE, text=True, stderr=sys.stderr,
            cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
    loop = asyncio.get_running_loop()
    dts = []
    for ch in procs:
        line = await loop.run_in_executor(None, ch.stdout.readline)
        dts.append(float(line.strip()))
        ch.wait(timeout=30)
    out[label] = round(k / max(dts), 1)


async def stat_worker(args):
    from curvine_amd.client.fs_client import FsClient
    from curvine_amd.conf import ClusterConf

    conf = ClusterConf()
    conf.client.master_addrs = [args.master]
    cl = FsClient(conf)
    rng = random.Random(args.seed)
    idxs = [rng.randrange(args.lo, args.n) for _ in range(args.k)]
    conc = 10
    t0 = time.perf_counter()

    async def w(off):
        for j in range(off, len(idxs), conc):
            i = idxs[j]
            await cl.file_status(f"/pg/d{i // 1000}/f{i}")
    await asyncio.gather(*[w(off) for off in range(conc)])
    print(f"{time.perf_counter() - t0:.6f}", flush=True)
    await cl.close()


async def main_async(args):
    from curvine_amd.master.server import Master
    from curvine_amd.testing import test_conf

    results = {"n": args.n, "resident_cap": args.resident}
    for paged in (True, False) if not args.paged_only else (True,):
        tmp = tempfile.mkdtemp(prefix="paged-meta-")
        conf = test_conf(tmp)
        conf.master.max_resident_inodes = args.resident if paged else 0
        m = await Master(conf).start()
        sub = {}
        await build(m, args.n, sub)
        # hot working set FITS the resident cap (half of it, random);
        # cold touches the whole namespace (fault-in path)
        hot_n = min(args.n, max(1000, args.resident // 2))
        await stat_qps(m.rpc.port, hot_n, args.k, 7, "stat_qps_pass1", sub)
        await stat_qps(m.rpc.port, hot_n, args.k, 8, "stat_qps_hot", sub)
        await stat_qps(m.rpc.port, args.n, args.k, 99, "stat_qps_cold", sub,
                       lo=0)
        sub["rss_mb_final"] = round(rss_mb(), 1)
        sub["resident_final"] = len(m.fs.fs_dir.inodes)
        await m.stop()
        results["paged" if paged else "unpaged"] = sub
        print(json.dumps({("paged" if paged else "unpaged"): sub}),
              flush=True)
    print(json.dumps(results))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=500_000)
    p.add_argument("--resident", type=int, default=50_000)
    p.add_argument("--k", type=int, default=20_000)
    p.add_argument("--paged-only", action="store_true")
    p.add_argument("--stat-worker", action="store_true")
    p.add_argument("--master", default="")
    p.add_argument("--seed", type=int, default=1)
    p.add_argument("--lo", type=int, default=0)
    args = p.parse_args()
    if args.stat_worker:
        asyncio.new_event_loop().run_until_complete(stat_worker(args))
    else:
        asyncio.new_event_loop().run_until_complete(main_async(args))


if __name__ == "__main__":
    main()
