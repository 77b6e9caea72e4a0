#!/usr/bin/env python3
"""Metadata QPS benchmark at fixed concurrency (the reference's headline
metadata table: create/open/rename/delete QPS at concurrency 40,
README.md:92-99 / BASELINE.md).

The master runs in this process; load is driven from ``--procs`` separate
client processes (each an asyncio loop with ``--concurrency/--procs``
workers) so the measurement is server capacity, not one client GIL.

Usage: python scripts/meta_bench.py [--n 20000] [--concurrency 40]
       [--procs 8] [--no-native]
"""
import argparse
import asyncio
import json
import os
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

PHASES = ["create", "open", "stat", "rename", "delete"]


# ---------------------------------------------------------------- worker

async def worker_main(args):
    """One client process: waits for 'GO <phase>' lines on stdin, runs its
    slice [lo, hi) of the key range, prints 'DONE <phase> <dt>'."""
    from curvine_amd.client.fs_client import FsClient
    from curvine_amd.conf import ClusterConf

    conf = ClusterConf()
    conf.client.master_addrs = [args.master]
    conf.client.rpc_timeout_ms = 120_000
    clients = [FsClient(conf) for _ in range(args.clients)]
    conc = args.concurrency
    lo, hi = args.lo, args.hi

    def cl(i):
        return clients[i % len(clients)]

    async def create(i):
        await cl(i).create(f"/bench/f{i}", overwrite=True)
        await cl(i).complete_file(f"/bench/f{i}", 0, [])

    async def open_(i):
        await cl(i).open(f"/bench/f{i}")

    async def stat(i):
        await cl(i).file_status(f"/bench/f{i}")

    async def rename(i):
        await cl(i).rename(f"/bench/f{i}", f"/bench/g{i}")

    async def delete(i):
        await cl(i).delete(f"/bench/g{i}")

    fns = {"create": create, "open": open_, "stat": stat,
           "rename": rename, "delete": delete}

    loop = asyncio.get_running_loop()
    reader = asyncio.StreamReader()
    await loop.connect_read_pipe(
        lambda: asyncio.StreamReaderProtocol(reader), sys.stdin)
    print("READY", flush=True)
    while True:
        line = (await reader.readline()).decode().strip()
        if not line or line == "QUIT":
            break
        phase = line.split()[1]
        fn = fns[phase]

        async def w(off):
            for i in range(lo + off, hi, conc):
                await fn(i)
        t0 = time.perf_counter()
        await asyncio.gather(*[w(off) for off in range(conc)])
        dt = time.perf_counter() - t0
        print(f"DONE {phase} {dt:.6f}", flush=True)
    for c in clients:
        await c.close()


# ---------------------------------------------------------------- driver

async def main_async(args):
    from curvine_amd.testing import MiniCluster, test_conf

    tmp = tempfile.mkdtemp(prefix="meta-bench-")
    conf = test_conf(tmp)
    if args.no_native:
        conf.master.native_meta = False
    async with MiniCluster(conf=conf, tmp_dir=tmp) as mc:
        master = f"127.0.0.1:{mc.master.rpc.port}"
        n, conc, procs = args.n, args.concurrency, args.procs
        per = max(1, conc // procs)
        results = {"n": n, "concurrency": per * procs, "procs": procs,
                   "native": not args.no_native}
        # spawn client processes over disjoint key slices
        children = []
        step = n // procs
        for p in range(procs):
            lo, hi = p * step, (p + 1) * step if p < procs - 1 else n
            cmd = [sys.executable, os.path.abspath(__file__), "--worker",
                   "--master", master, "--lo", str(lo), "--hi", str(hi),
                   "--concurrency", str(per), "--clients",
                   str(max(1, args.clients // procs))]
            children.append(subprocess.Popen(
                cmd, stdin=subprocess.PIPE, stdout=subprocess.PIPE,
                text=True, bufsize=1,
                cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__)))))
        loop = asyncio.get_running_loop()
        for ch in children:
            line = await loop.run_in_executor(None, ch.stdout.readline)
            assert line.strip() == "READY", line
        for phase in PHASES:
            for ch in children:
                ch.stdin.write(f"GO {phase}\n")
                ch.stdin.flush()
            dts = []
            for ch in children:
                line = await loop.run_in_executor(None, ch.stdout.readline)
                parts = line.split()
                assert parts[0] == "DONE", line
                dts.append(float(parts[2]))
            results[f"{phase}_qps"] = round(n / max(dts), 1)
        for ch in children:
            ch.stdin.write("QUIT\n")
            ch.stdin.flush()
        for ch in children:
            ch.wait(timeout=30)
        if mc.master.native_meta is not None:
            results["meta_stats"] = mc.master.native_meta.stats()
        print(json.dumps(results))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=20000)
    p.add_argument("--concurrency", type=int, default=40)
    p.add_argument("--clients", type=int, default=8)
    p.add_argument("--procs", type=int, default=8)
    p.add_argument("--no-native", action="store_true")
    # internal worker mode
    p.add_argument("--worker", action="store_true")
    p.add_argument("--master", default="")
    p.add_argument("--lo", type=int, default=0)
    p.add_argument("--hi", type=int, default=0)
    args = p.parse_args()
    if args.worker:
        asyncio.new_event_loop().run_until_complete(worker_main(args))
    else:
        asyncio.new_event_loop().run_until_complete(main_async(args))


if __name__ == "__main__":
    main()
