#!/usr/bin/env python3
"""Kernel/data-path microbenchmarks on the MI355X: arena H2D/D2H bandwidth,
CRC32C kernel throughput, gather (extent-pack) throughput, fill, D2D.

Run under rocprofv3 for per-kernel stats:
    rocprofv3 --kernel-trace --stats -d gpurun_out/prof -- \
        python scripts/gpu_microbench.py
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
from curvine_amd import native


def timeit(fn, n=5, warmup=2):
    for _ in range(warmup):
        fn()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    return (time.perf_counter() - t0) / n


def main():
    assert native.gpu_available(), "needs a GPU"
    GB = 1 << 30
    size = 4 * GB
    a = native.Arena(0, size + (1 << 20), staging_bytes=16 << 20,
                     staging_count=8)
    results = {}

    host = np.random.default_rng(0).integers(0, 256, 1 * GB, dtype=np.uint8)
    pin = native.PinnedBuffer(1 * GB)

    # H2D via staging ring (pageable source)
    dt = timeit(lambda: a.write(0, host, 0, len(host)))
    results["h2d_pageable_ring_GBps"] = round(len(host) / dt / 1e9, 2)

    # H2D from pinned (direct DMA)
    pin.view[:len(host)] = host.tobytes()
    dt = timeit(lambda: a.write_from_ptr(0, pin.ptr, len(host), False))
    results["h2d_pinned_GBps"] = round(len(host) / dt / 1e9, 2)

    # D2H via staging ring into pageable
    out = np.zeros(1 * GB, dtype=np.uint8)
    dt = timeit(lambda: a.read(0, out, 0, len(out)))
    results["d2h_pageable_ring_GBps"] = round(len(out) / dt / 1e9, 2)

    # D2H direct into pinned
    dt = timeit(lambda: a.read_to_ptr(0, pin.ptr, len(host), False))
    results["d2h_pinned_GBps"] = round(len(host) / dt / 1e9, 2)
    assert bytes(pin.view[:1 << 20]) == host[:1 << 20].tobytes()

    # device CRC32C kernel (1 GiB resident)
    crc_host = native.crc32c(host)
    dt = timeit(lambda: a.crc32c(0, len(host)))
    assert a.crc32c(0, len(host)) == crc_host
    results["crc32c_device_GBps"] = round(len(host) / dt / 1e9, 2)
    results["crc32c_matches_host"] = True

    # host CRC32C (SSE4.2) for comparison
    dt = timeit(lambda: native.crc32c(host))
    results["crc32c_host_sse42_GBps"] = round(len(host) / dt / 1e9, 2)

    # gather: pack 256 x 1 MiB scattered extents into pinned
    rng = np.random.default_rng(1)
    offs = sorted(rng.choice(range(0, 3 * GB // (1 << 20)), 256,
                             replace=False).tolist())
    extents = [(int(o) << 20, 1 << 20) for o in offs]
    total = sum(e[1] for e in extents)
    gout = np.zeros(total, dtype=np.uint8)
    dt = timeit(lambda: a.gather(extents, gout, 0))
    results["gather_256x1MiB_GBps"] = round(total / dt / 1e9, 2)

    # device fill
    dt = timeit(lambda: a.fill(0, 1 * GB, 0))
    results["fill_device_GBps"] = round(GB / dt / 1e9, 2)

    # D2D copy within arena (block move / promotion analog)
    dt = timeit(lambda: native.load().arena_copy(a.handle, 2 * GB, a.handle, 0, GB))
    results["d2d_copy_GBps"] = round(2 * GB / dt / 1e9, 2)  # rd+wr bytes

    # on-device sample gather (CurvineDeviceLoader hot path): 4096 x
    # 256 KiB scattered extents packed to a device destination
    mod = native.load()
    n_s, s_sz = 4096, 256 << 10
    src_offs = rng.choice(range(0, (2 * GB) // s_sz), n_s,
                          replace=False).tolist()
    triples = [(int(o) * s_sz, i * s_sz, s_sz)
               for i, o in enumerate(src_offs)]
    dst_ptr = a.base_ptr() + 3 * GB
    dt = timeit(lambda: mod.arena_gather_ptr(a.handle, triples, dst_ptr))
    results["gather_dev_4096x256KiB_GBps"] = round(
        2 * n_s * s_sz / dt / 1e9, 2)   # rd+wr bytes

    # LZ4 device decompress: 512 MiB of ~3:1-compressible data
    rep = np.frombuffer((b"curvine-amd lz4 block payload %06d " % 42) * 64,
                        dtype=np.uint8)
    block = np.tile(rep, (512 << 20) // len(rep) + 1)[:512 << 20].copy()
    noise_idx = rng.choice(len(block), len(block) // 64, replace=False)
    block[noise_idx] = rng.integers(0, 256, len(noise_idx), dtype=np.uint8)
    comp = native.lz4_compress(block.tobytes())
    results["lz4_ratio"] = round(len(block) / len(comp), 2)
    dt = timeit(lambda: mod.arena_lz4_decompress(a.handle, 0, comp))
    results["lz4_device_decompress_GBps"] = round(len(block) / dt / 1e9, 2)
    back = np.zeros(1 << 20, dtype=np.uint8)
    a.read(0, back, 0, 1 << 20)
    assert back.tobytes() == block[:1 << 20].tobytes()
    dt = timeit(lambda: native.lz4_decompress(comp))
    results["lz4_host_decompress_GBps"] = round(len(block) / dt / 1e9, 2)

    print(json.dumps(results, indent=1))
    a.close()
    pin.close()


if __name__ == "__main__":
    main()
