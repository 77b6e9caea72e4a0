#!/usr/bin/env python3
"""Flagship benchmark: cached sequential-read GB/s on MI355X HBM tiers.

Measures the BASELINE.json metric (FUSE/client sequential-read throughput
on synthetic random-byte files cached in HBM) on N GPUs of one node —
weak scaling: each rank owns one GPU's worker (HBM arena) and reads its
own shard through the cache client (short-circuit HBM -> pinned -> host),
fio-style with T concurrent streams of 1 MiB reads.

Driver contract: rank 0 prints ONE JSON line; timing brackets are
barrier + torch.cuda.synchronize on both sides; value is the whole-job
aggregate GiB/s (max elapsed over ranks).
"""
from __future__ import annotations

import argparse
import asyncio
import json
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

REFERENCE_PEAK_GIBPS = 9.5   # 256KB seq read, 32 threads


def log(rank, *a):
    print(f"[bench r{rank}]", *a, file=sys.stderr, flush=True)


async def setup_cluster(args, rank, world, dist):
    """Rank 0: master. Every rank: one worker on its GPU + client."""
    from curvine_amd.testing import test_conf
    from curvine_amd.master.server import Master
    from curvine_amd.worker.server import Worker
    from curvine_amd.client.filesystem import CurvineFileSystem

    tmp = tempfile.mkdtemp(prefix=f"curvine-bench-r{rank}-")
    conf = test_conf(tmp)
    conf.master.block_size = args.block_size
    conf.client.block_size = args.block_size
    conf.master.heartbeat_check_ms = 1000
    conf.worker.heartbeat_interval_ms = 1000
    conf.client.write_chunk_size = 4 << 20
    conf.client.read_chunk_size = args.read_chunk
    conf.worker.staging_buf_bytes = args.staging_bytes
    conf.worker.staging_buf_count = args.staging_count

    master = None
    if rank == 0:
        conf.master.rpc_port = 0
        master = await Master(conf).start()
        master_port = master.rpc.port
    else:
        master_port = 0
    if world > 1:
        obj = [master_port]
        dist.broadcast_object_list(obj, src=0)
        master_port = obj[0]
    conf.client.master_addrs = [f"127.0.0.1:{master_port}"]
    conf.master.rpc_port = master_port

    local_rank = int(os.environ.get("LOCAL_RANK", rank))
    from curvine_amd import native
    hbm_gb = args.hbm_gb
    if native.gpu_available():
        conf.worker.data_dirs = [f"[HBM:{hbm_gb}GB:{local_rank}]gpu{local_rank}"]
    else:
        # CPU fallback for plumbing runs in the dev container
        conf.worker.data_dirs = [f"[MEM:{hbm_gb}GB]{tmp}/mem"]
    conf.worker.rpc_port = 0
    worker = await Worker(conf, worker_id=rank + 1, device_id=local_rank).start()

    fs = CurvineFileSystem(conf)
    fs.client.local_worker_id = worker.worker_id
    return master, worker, fs, conf


async def write_dataset(args, rank, fs):
    """Synthetic random-byte files, one directory per rank."""
    import numpy as np
    rng = np.random.default_rng(1234 + rank)
    base_chunk = rng.integers(0, 256, size=8 << 20, dtype=np.uint8).tobytes()
    for i in range(args.files):
        w = await fs.create(f"/bench/r{rank}/f{i}", overwrite=True,
                            storage_tier="HBM")
        remaining = args.file_size
        while remaining > 0:
            n = min(len(base_chunk), remaining)
            await w.write(base_chunk[:n] if n < len(base_chunk) else base_chunk)
            remaining -= n
        await w.complete()


async def one_step(args, rank, fs) -> int:
    """Read the whole rank-local dataset with T concurrent streams of
    `read_chunk` sequential reads. Returns bytes read."""
    total = 0
    sem = asyncio.Semaphore(args.threads)
    results = []

    async def read_file(i):
        async with sem:
            r = await fs.open(f"/bench/r{rank}/f{i}")
            got, pos = 0, 0
            buf = bytearray(args.read_chunk)
            while pos < r.length:
                n = await r.pread_into(pos, buf, 0, min(args.read_chunk,
                                                        r.length - pos))
                if n <= 0:
                    break
                pos += n
                got += n
            r.close()
            return got

    results = await asyncio.gather(*[read_file(i) for i in range(args.files)])
    total = sum(results)
    expect = args.files * args.file_size
    if total != expect:
        raise RuntimeError(f"step read {total} != {expect}")
    return total


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=5)
    p.add_argument("--warmup", type=int, default=2)
    p.add_argument("--files", type=int, default=8)
    p.add_argument("--file-size", type=int, default=1 << 30)
    p.add_argument("--block-size", type=int, default=256 << 20)
    p.add_argument("--read-chunk", type=int, default=1 << 20)
    p.add_argument("--threads", type=int, default=8)
    p.add_argument("--hbm-gb", type=int, default=16)
    p.add_argument("--staging-bytes", type=int, default=8 << 20)
    p.add_argument("--staging-count", type=int, default=8)
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))

    import torch
    dist = None
    if world > 1:
        import torch.distributed as dist_mod
        dist = dist_mod
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
        if torch.cuda.is_available():
            torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", rank)))

    has_gpu = torch.cuda.is_available()
    # size the dataset to the machine: CPU plumbing runs use small files
    if not has_gpu and args.file_size > 64 << 20:
        args.files, args.file_size = 4, 32 << 20
        args.hbm_gb = 1

    def barrier_sync():
        if dist is not None:
            dist.barrier()
        if has_gpu:
            torch.cuda.synchronize()

    loop = asyncio.new_event_loop()
    asyncio.set_event_loop(loop)

    master, worker, fs, conf = loop.run_until_complete(
        setup_cluster(args, rank, world, dist))
    log(rank, f"cluster up; writing {args.files}x{args.file_size >> 20}MiB")
    t0 = time.perf_counter()
    loop.run_until_complete(write_dataset(args, rank, fs))
    log(rank, f"dataset written in {time.perf_counter() - t0:.1f}s")

    for _ in range(args.warmup):
        loop.run_until_complete(one_step(args, rank, fs))

    barrier_sync()
    t_start = time.perf_counter()
    bytes_per_step = 0
    for _ in range(args.steps):
        bytes_per_step = loop.run_until_complete(one_step(args, rank, fs))
    barrier_sync()
    elapsed = time.perf_counter() - t_start

    # max elapsed over ranks
    if dist is not None:
        t = torch.tensor([elapsed], dtype=torch.float64)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = t.item()

    total_bytes = bytes_per_step * args.steps * world
    gibps = total_bytes / elapsed / (1 << 30)

    if rank == 0:
        result = {
            "metric": "cached_seq_read_GiBps",
            "value": round(gibps, 3),
            "unit": "GiB/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": round(elapsed / args.steps * 1000, 2),
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": round(gibps / REFERENCE_PEAK_GIBPS, 3),
            "dtype": "bytes",
            "data": "synthetic random-byte files (numpy PRNG), cached in HBM tier",
            "config": {
                "model": "curvine-amd cache engine",
                "workload": "fio-style sequential read, cached",
                "path": "client_short_circuit",
                "tier": "HBM" if has_gpu else "MEM(cpu plumbing)",
                "files_per_rank": args.files,
                "file_size": args.file_size,
                "read_chunk": args.read_chunk,
                "threads": args.threads,
                "block_size": args.block_size,
                "parallelism": f"shard-per-gpu x{world}",
            },
        }
        print(json.dumps(result), flush=True)

    async def teardown():
        await fs.close()
        await worker.stop()
        if master:
            await master.stop()
    loop.run_until_complete(teardown())
    if dist is not None:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
