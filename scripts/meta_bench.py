#!/usr/bin/env python3
"""Metadata QPS benchmark at fixed concurrency (the reference's headline
metadata table: create/open/rename/delete QPS at concurrency 40,
README.md:92-99 / BASELINE.md).

Usage: python scripts/meta_bench.py [--n 20000] [--concurrency 40]
"""
import argparse
import asyncio
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


async def timed(name, n, conc, fn):
    """fn(i) -> coroutine; n ops across `conc` workers."""
    t0 = time.perf_counter()

    async def worker(w):
        for i in range(w, n, conc):
            await fn(i)
    await asyncio.gather(*[worker(w) for w in range(conc)])
    dt = time.perf_counter() - t0
    return round(n / dt, 1)


async def main_async(args):
    from curvine_amd.client.fs_client import FsClient
    from curvine_amd.testing import MiniCluster

    tmp = tempfile.mkdtemp(prefix="meta-bench-")
    async with MiniCluster(tmp_dir=tmp) as mc:
        conf = mc.client_conf()
        conf.client.rpc_timeout_ms = 120_000
        # several client connections share the load (concurrency 40 on a
        # handful of sockets, like the reference's bench)
        clients = [FsClient(conf) for _ in range(args.clients)]
        n, conc = args.n, args.concurrency
        results = {"n": n, "concurrency": conc}

        def cl(i):
            return clients[i % len(clients)]

        async def create(i):
            await cl(i).create(f"/bench/f{i}", overwrite=True)
            await cl(i).complete_file(f"/bench/f{i}", 0, [])
        results["create_qps"] = await timed("create", n, conc, create)

        async def open_(i):
            await cl(i).open(f"/bench/f{i}")
        results["open_qps"] = await timed("open", n, conc, open_)

        async def stat(i):
            await cl(i).file_status(f"/bench/f{i}")
        results["stat_qps"] = await timed("stat", n, conc, stat)

        async def rename(i):
            await cl(i).rename(f"/bench/f{i}", f"/bench/g{i}")
        results["rename_qps"] = await timed("rename", n, conc, rename)

        async def delete(i):
            await cl(i).delete(f"/bench/g{i}")
        results["delete_qps"] = await timed("delete", n, conc, delete)

        for c in clients:
            await c.close()
        print(json.dumps(results))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--n", type=int, default=20000)
    p.add_argument("--concurrency", type=int, default=40)
    p.add_argument("--clients", type=int, default=8)
    args = p.parse_args()
    asyncio.new_event_loop().run_until_complete(main_async(args))


if __name__ == "__main__":
    main()
