#!/usr/bin/env python3
"""Training-data path benchmark (BASELINE config[4] analog): WebDataset-
style tar shards cached in the HBM tier, consumed by a torch DataLoader;
sustained sample GB/s reported.  Single GPU (the driver's 8-GPU tier runs
bench.py; this script covers the DataLoader integration path).

    python scripts/dataloader_bench.py [--shards 16] [--shard-mb 512]
"""
import argparse
import io
import json
import os
import sys
import tarfile
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--shards", type=int, default=16)
    p.add_argument("--shard-mb", type=int, default=512)
    p.add_argument("--sample-kb", type=int, default=256)
    p.add_argument("--epochs", type=int, default=3)
    p.add_argument("--workers", type=int, default=0,
                   help="DataLoader num_workers (forked workers read via "
                        "RPC; in-process short-circuit is parent-only)")
    p.add_argument("--to-device", action="store_true",
                   help="land each sample in a cuda tensor")
    p.add_argument("--device-loader", action="store_true",
                   help="CurvineDeviceLoader: batched HBM->device gather "
                        "(no host hop) instead of torch DataLoader")
    p.add_argument("--batch-size", type=int, default=64)
    args = p.parse_args()

    import numpy as np
    import torch
    from torch.utils.data import DataLoader

    from curvine_amd import native
    from curvine_amd.client.filesystem import SyncFs
    from curvine_amd.sdk.dataset import CurvineShardDataset
    from curvine_amd.testing import SyncMiniCluster, test_conf

    has_gpu = native.gpu_available()
    if not has_gpu:
        args.shards, args.shard_mb = 4, 64
    tmp = tempfile.mkdtemp(prefix="dl-bench-")
    conf = test_conf(tmp)
    conf.master.block_size = 256 << 20
    conf.client.block_size = 256 << 20
    hbm = args.shards * args.shard_mb // 1024 + 4
    smc = SyncMiniCluster(
        conf=conf, tmp_dir=tmp,
        worker_dirs=[[f"[HBM:{hbm}GB:0]gpu0" if has_gpu
                      else f"[MEM:2GB]{tmp}/mem"]]).start()
    cconf = smc.client_conf()
    sf = SyncFs(cconf)
    sf.fs.client.local_worker_id = smc.workers[0].worker_id

    # build + cache shards
    rng = np.random.default_rng(0)
    sample = rng.integers(0, 256, args.sample_kb << 10, dtype=np.uint8).tobytes()
    per_shard = (args.shard_mb << 20) // len(sample)
    t0 = time.perf_counter()
    shard_paths = []
    for s in range(args.shards):
        buf = io.BytesIO()
        with tarfile.open(fileobj=buf, mode="w") as tf:
            for i in range(per_shard):
                info = tarfile.TarInfo(f"s{s:03d}/{i:06d}.bin")
                info.size = len(sample)
                tf.addfile(info, io.BytesIO(sample))
        path = f"/shards/shard-{s:04d}.tar"
        sf.write_file(path, buf.getvalue(), storage_tier="HBM")
        shard_paths.append(path)
    ingest_s = time.perf_counter() - t0
    total_mb = args.shards * args.shard_mb

    if args.device_loader:
        from curvine_amd.sdk.dataset import CurvineDeviceLoader
        dl = CurvineDeviceLoader(
            cconf, shard_paths,
            device="cuda:0" if has_gpu else "cpu",
            batch_size=args.batch_size)
        results = {"shards": args.shards, "shard_mb": args.shard_mb,
                   "sample_kb": args.sample_kb, "mode": "device_loader",
                   "samples": dl.num_samples,
                   "ingest_GBps": round(total_mb / 1024 / ingest_s, 2),
                   "tier": "HBM" if has_gpu else "MEM"}
        epochs = []
        for _ in range(args.epochs):
            n = 0
            t0 = time.perf_counter()
            for tensor, sections, names in dl:
                n += tensor.numel()
            if has_gpu:
                torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            epochs.append(round(n / dt / 2**30, 3))
        results["epoch_GiBps"] = epochs
        results["sustained_GiBps"] = max(epochs)
        print(json.dumps(results))
        dl.close()
        sf.shutdown()
        smc.stop()
        return

    dev = torch.device("cuda:0") if has_gpu and args.to_device else None
    # device transfer must happen in the PARENT when num_workers > 0
    # (forked children cannot touch the HIP context)
    move_in_transform = dev is not None and args.workers == 0

    def xform(item):
        name, payload = item
        if move_in_transform:
            return torch.frombuffer(bytearray(payload), dtype=torch.uint8) \
                .to(dev, non_blocking=True)
        return payload

    ds = CurvineShardDataset(cconf, shard_paths, transform=xform)
    loader = DataLoader(ds, batch_size=None, num_workers=args.workers)

    results = {"shards": args.shards, "shard_mb": args.shard_mb,
               "sample_kb": args.sample_kb,
               "ingest_GBps": round(total_mb / 1024 / ingest_s, 2),
               "tier": "HBM" if has_gpu else "MEM"}
    epochs = []
    for e in range(args.epochs):
        n = 0
        t0 = time.perf_counter()
        for item in loader:
            if dev is not None and not move_in_transform:
                item = torch.frombuffer(bytearray(item), dtype=torch.uint8) \
                    .to(dev, non_blocking=True)
            n += item.numel() if hasattr(item, "numel") else len(item)
        if has_gpu:
            torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        epochs.append(round(n / dt / 2**30, 3))
    results["epoch_GiBps"] = epochs
    results["sustained_GiBps"] = max(epochs)
    results["to_device"] = bool(dev is not None)
    print(json.dumps(results))

    sf.shutdown()
    smc.stop()


if __name__ == "__main__":
    main()
