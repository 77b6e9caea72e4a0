"""Torch-facing device reads: cached file bytes land directly in GPU
tensors.

The "client reads land directly in consumer GPU memory" path of
BASELINE.json: when the file's blocks sit in the local worker's HBM arena,
`read_into_tensor` is a device-to-device hipMemcpyAsync (no host round
trip); host-tier blocks are uploaded through the pinned path.
"""
from __future__ import annotations


from curvine_amd import errors as err


def _block_readers(sync_reader):
    return sync_reader._readers


class CurvineTensorReader:
    """Reusable reader for one cached file, filling torch tensors."""

    def __init__(self, sync_fs, path: str):
        self.sync_fs = sync_fs
        r = sync_fs.call(sync_fs.fs.open(path))
        try:
            self.reader = r.to_sync()
        finally:
            r.close()
        self.length = self.reader.length

    def read_into_tensor(self, tensor, file_off: int = 0,
                         n: int | None = None) -> int:
        """Fill `tensor` (uint8, contiguous; cpu or cuda) with file bytes.
        Returns bytes read."""
        import torch
        assert tensor.dtype == torch.uint8 and tensor.is_contiguous()
        cap = tensor.numel()
        n = cap if n is None else min(n, cap)
        n = max(0, min(n, self.length - file_off))
        if n == 0:
            return 0
        is_dev = tensor.is_cuda
        got = 0
        import bisect
        offs = self.reader._offs
        fb = self.reader.fb
        while got < n:
            idx = bisect.bisect_right(offs, file_off + got) - 1
            lb = fb.blocks[idx]
            boff = file_off + got - lb.offset
            want = min(n - got, lb.block.length - boff)
            if want <= 0:
                break
            got += self.reader._readers[idx].read_to_ptr(
                boff, tensor.data_ptr() + got, want, is_dev)
        if is_dev:
            torch.cuda.synchronize(tensor.device)
        return got

    def to_tensor(self, device="cuda:0", file_off: int = 0,
                  n: int | None = None):
        import torch
        n = self.length - file_off if n is None else n
        t = torch.empty(n, dtype=torch.uint8, device=device)
        got = self.read_into_tensor(t, file_off, n)
        return t[:got]

    def close(self):
        self.reader.close()


def read_into_tensor(sync_fs, path: str, tensor, file_off: int = 0) -> int:
    """One-shot helper: curvine path -> torch tensor (device or host)."""
    r = CurvineTensorReader(sync_fs, path)
    try:
        return r.read_into_tensor(tensor, file_off)
    finally:
        r.close()
