#!/usr/bin/env python3
"""FUSE throughput diagnostic: mounts a single-GPU cache, writes files,
then measures reads with external `dd` processes (no Python reader GIL)
and prints the daemon's per-op stats.

Usage: python scripts/fuse_probe.py [--files N] [--file-size BYTES]
"""
import argparse
import json
import os
import signal
import subprocess
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--files", type=int, default=8)
    p.add_argument("--file-size", type=int, default=1 << 30)
    p.add_argument("--channels", type=int, default=16)
    p.add_argument("--bs", default="1M")
    args = p.parse_args()

    from curvine_amd import native
    from curvine_amd.testing import SyncMiniCluster, test_conf

    has_gpu = native.gpu_available()
    if not os.path.exists("/dev/fuse"):
        os.mknod("/dev/fuse", 0o666 | 0o020000, os.makedev(10, 229))
    tmp = tempfile.mkdtemp(prefix="fuse-probe-")
    conf = test_conf(tmp)
    conf.master.block_size = 256 << 20
    smc = SyncMiniCluster.__new__(SyncMiniCluster)
    import asyncio
    import threading
    smc.loop = asyncio.new_event_loop()
    smc._thread = threading.Thread(
        target=lambda: (asyncio.set_event_loop(smc.loop), smc.loop.run_forever()),
        daemon=True)
    smc._thread.start()
    from curvine_amd.testing import MiniCluster
    smc.mc = MiniCluster(conf=conf, tmp_dir=tmp, workers=0)
    smc.call(smc.mc.start())
    mnt = "/tmp/fuse-probe-mnt"
    dirs = [f"[HBM:{args.files * (args.file_size >> 30) + 4}GB:0]gpu0"] \
        if has_gpu else [f"[MEM:4GB]{tmp}/mem"]
    if not has_gpu:
        args.file_size = min(args.file_size, 256 << 20)
    daemon = subprocess.Popen(
        [sys.executable, "-m", "curvine_amd.fuse", "--mnt", mnt,
         "--master", f"127.0.0.1:{smc.mc.master.rpc.port}",
         "--embed-worker", "--device", "0" if has_gpu else "-1",
         "--channels", str(args.channels), "--log-level", "WARNING"] +
        [f"--data-dir={d}" for d in dirs],
        stdout=subprocess.PIPE, stderr=None, text=True,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert daemon.stdout.readline().startswith("READY")

    # write dataset through the mount with dd from /dev/urandom? dd from
    # urandom is slow; write from a prebuilt local file
    seed = os.path.join(tmp, "seed.bin")
    with open(seed, "wb") as f:
        f.write(os.urandom(64 << 20))
    t0 = time.perf_counter()
    for i in range(args.files):
        with open(seed, "rb") as src, open(f"{mnt}/f{i}", "wb") as dst:
            remaining = args.file_size
            while remaining > 0:
                src.seek(0)
                n = min(64 << 20, remaining)
                dst.write(src.read(n))
                remaining -= n
    wt = time.perf_counter() - t0
    total = args.files * args.file_size
    print(f"WRITE: {total / wt / 2**30:.2f} GiB/s ({wt:.1f}s)")

    def run_dd(n_procs, drop_cache=True):
        if drop_cache:
            with open("/proc/sys/vm/drop_caches", "w") as f:
                f.write("3")
        procs = []
        t0 = time.perf_counter()
        for i in range(n_procs):
            procs.append(subprocess.Popen(
                ["dd", f"if={mnt}/f{i % args.files}", "of=/dev/null",
                 f"bs={args.bs}"],
                stdout=subprocess.DEVNULL, stderr=subprocess.DEVNULL))
        for pr in procs:
            pr.wait()
        dt = time.perf_counter() - t0
        return n_procs * args.file_size / dt / 2**30

    for n in (1, 4, 8, 16):
        gibps = run_dd(n)
        print(f"dd x{n} bs={args.bs}: {gibps:.2f} GiB/s")

    daemon.send_signal(signal.SIGUSR1)
    time.sleep(0.5)
    daemon.terminate()
    daemon.wait(timeout=15)
    smc.call(smc.mc.stop())
    smc.loop.call_soon_threadsafe(smc.loop.stop)
    print("DONE")


if __name__ == "__main__":
    main()
