"""Block distribution over torch.distributed collectives.

CPU: gloo, world_size=2, host arenas (multi-process via mp.spawn).
GPU: DLPack zero-copy interop between HBM arenas and torch tensors.
"""
import os

import numpy as np
import pytest


def _dist_worker(rank, world, port, tmp_dir):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        from curvine_amd.conf import WorkerConf
        from curvine_amd.parallel import BlockDistributor
        from curvine_amd.worker.block_store import BlockStore

        conf = WorkerConf(data_dirs=[f"[MEM:64MB]{tmp_dir}/r{rank}"])
        store = BlockStore(conf)
        length = 10 << 20
        payload = np.random.default_rng(7).integers(
            0, 256, length, dtype=np.uint8).tobytes()
        if rank == 0:
            w = store.create_writer(42, length, "MEM")
            w.write(payload)
            store.finalize(42, length)
        bd = BlockDistributor()
        bd.broadcast_block(store, 42, length, src_rank=0, tier="MEM",
                           chunk=3 << 20)
        r = store.open_reader(42)
        got = r.read(0, length)
        r.close()
        assert got == payload, f"rank {rank} mismatch"
        # extent collective: allgather shards
        from curvine_amd.native import Arena
        a = Arena(-1, 1 << 20)
        n = 1024
        mine_off = rank * n
        a.write(mine_off, bytes([rank + 1]) * n)
        bd.allgather_extents(a, mine_off, [0, n], n)
        assert a.read_bytes(0, n) == bytes([1]) * n
        assert a.read_bytes(n, n) == bytes([2]) * n
        a.close()
        store.close()
        dist.barrier()
    finally:
        dist.destroy_process_group()


def test_broadcast_block_gloo_2proc(tmp_path):
    import socket

    import torch.multiprocessing as mp
    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    mp.spawn(_dist_worker, args=(2, port, str(tmp_path)), nprocs=2,
             join=True)


def test_arena_tensor_cpu():
    import torch

    from curvine_amd.native import Arena
    from curvine_amd.parallel.distributor import arena_tensor
    a = Arena(-1, 1 << 20)
    a.write(100, b"\x05" * 1000)
    t = arena_tensor(a, 100, 1000)
    assert t.dtype == torch.uint8 and t.shape == (1000,)
    assert int(t.sum()) == 5 * 1000
    # zero copy: mutating the tensor mutates the arena
    t[:10] = 9
    assert a.read_bytes(100, 10) == b"\x09" * 10
    a.close()


@pytest.mark.gpu
def test_arena_tensor_hbm():
    import torch

    from curvine_amd.native import Arena
    from curvine_amd.parallel.distributor import arena_tensor
    a = Arena(0, 64 << 20)
    data = np.random.default_rng(3).integers(0, 256, 1 << 20, dtype=np.uint8)
    a.write(4096, data, 0, len(data))
    t = arena_tensor(a, 4096, len(data))
    assert t.device.type == "cuda"
    assert int(t.sum()) == int(data.astype(np.uint64).sum())
    # writes through torch land in the arena
    t.zero_()
    torch.cuda.synchronize()
    assert a.read_bytes(4096, 16) == b"\x00" * 16
    a.close()
