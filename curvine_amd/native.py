"""Loader/wrapper for the native HIP data plane (csrc/ -> _native.so).

Policy: on a machine WITH a GPU the native extension is mandatory — ops
fail loudly rather than falling back to a silent CPU path.  On CPU-only
machines the same extension is still used (its host-memory arena code), so
CPU tests exercise the identical C++ code paths.

If the .so is missing, we build it in-tree with hipcc (gfx950 cross-compile
works without a GPU; ~15 s cold).
"""
from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
import threading

_REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
_SO = os.path.join(_REPO, "curvine_amd", "_native.so")
_SRC = os.path.join(_REPO, "csrc", "module.cpp")
_SRC2 = os.path.join(_REPO, "csrc", "kernels.hip")
_SRC3 = os.path.join(_REPO, "csrc", "lz4.hip")
_SRC4 = os.path.join(_REPO, "csrc", "fuse_loop.hip")
_SRC5 = os.path.join(_REPO, "csrc", "meta_server.cpp")
_SRC6 = os.path.join(_REPO, "csrc", "data_server.cpp")
_SRC7 = os.path.join(_REPO, "csrc", "sdk_abi.cpp")
_lock = threading.Lock()
_mod = None


def build_native(force: bool = False) -> str:
    """Compile csrc/ into curvine_amd/_native.so for gfx950."""
    if not force and os.path.exists(_SO) and os.path.exists(_SRC):
        if os.path.getmtime(_SO) >= max(os.path.getmtime(_SRC),
                                        os.path.getmtime(_SRC2),
                                        os.path.getmtime(_SRC3),
                                        os.path.getmtime(_SRC4),
                                        os.path.getmtime(_SRC5),
                                        os.path.getmtime(_SRC6),
                                        os.path.getmtime(_SRC7)):
            return _SO
    import pybind11
    hipcc = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")
    cmd = [hipcc, "--offload-arch=gfx950", "-O3", "-std=c++17", "-fPIC",
           "-shared", "-msse4.2",
           f"-I{pybind11.get_include()}",
           f"-I{sysconfig.get_paths()['include']}",
           _SRC, "-o", _SO]
    subprocess.run(cmd, check=True, capture_output=True, text=True)
    return _SO


def load():
    global _mod
    if _mod is not None:
        return _mod
    with _lock:
        if _mod is not None:
            return _mod
        # Load torch's bundled HIP runtime first: loading /opt/rocm's
        # libamdhip64 ahead of torch breaks torch.cuda init ("No HIP GPUs
        # are available") — observed on the MI355X pool (torch 2.10+rocm7.0
        # vs /opt/rocm 7.2).  Harmless when torch is absent.
        try:
            import torch  # noqa: F401
        except ImportError:
            pass
        try:
            from curvine_amd import _native as mod  # type: ignore
        except ImportError:
            build_native()
            import importlib
            import curvine_amd
            importlib.invalidate_caches()
            from curvine_amd import _native as mod  # type: ignore
        _mod = mod
        return mod


def gpu_available() -> bool:
    try:
        return load().device_count() > 0
    except Exception:
        return False


def device_count() -> int:
    return load().device_count()


class Arena:
    """A contiguous byte arena: HBM (device >= 0) or host memory
    (device == -1).  Byte movement only; offset allocation is the caller's
    (see curvine_amd.worker.arena_alloc)."""

    def __init__(self, device: int, capacity: int,
                 staging_bytes: int = 4 << 20, staging_count: int = 8,
                 host_pinned: bool = False):
        self._n = load()
        self.device = device
        self.capacity = capacity
        self.handle = self._n.arena_create(device, capacity, staging_bytes,
                                           staging_count, host_pinned)
        self._closed = False

    # ---- byte movement ----
    def write(self, off: int, buf, buf_off: int = 0, n: int | None = None) -> None:
        if n is None:
            n = len(buf) - buf_off
        self._n.arena_write(self.handle, off, buf, buf_off, n)

    def read(self, off: int, out, out_off: int = 0, n: int | None = None) -> None:
        if n is None:
            n = len(out) - out_off
        self._n.arena_read(self.handle, off, out, out_off, n)

    def read_bytes(self, off: int, n: int) -> bytes:
        out = bytearray(n)
        self._n.arena_read(self.handle, off, out, 0, n)
        return bytes(out)

    def read_to_ptr(self, off: int, dst_ptr: int, n: int, device: bool) -> None:
        self._n.arena_read_ptr(self.handle, off, dst_ptr, n, device)

    def write_from_ptr(self, off: int, src_ptr: int, n: int, device: bool) -> None:
        self._n.arena_write_ptr(self.handle, off, src_ptr, n, device)

    def copy_to(self, dst: "Arena", dst_off: int, src_off: int, n: int) -> None:
        self._n.arena_copy(dst.handle, dst_off, self.handle, src_off, n)

    def fill(self, off: int, n: int, value: int = 0) -> None:
        self._n.arena_fill(self.handle, off, n, value)

    def crc32c(self, off: int, n: int) -> int:
        return self._n.arena_crc32c(self.handle, off, n)

    def gather(self, extents: list[tuple[int, int]], out, out_off: int = 0) -> None:
        """Pack [(off, len)...] into `out` contiguously (GPU kernel +
        pipelined D2H on device arenas)."""
        self._n.arena_gather(self.handle, extents, out, out_off)

    def gather_ptr(self, triples: list[tuple[int, int, int]],
                   dst_ptr: int) -> None:
        """Scatter-gather [(src_off, dst_off, len)...] into a raw pointer.
        Device arena requires a device dst (on-chip D2D kernel, no host
        hop); host arena requires a host dst (memcpy)."""
        self._n.arena_gather_ptr(self.handle, triples, dst_ptr)

    def base_ptr(self) -> int:
        return self._n.arena_base_ptr(self.handle)

    def ipc_handle(self) -> bytes:
        """hipIpc export of a device arena (64-byte handle another
        process opens with Arena.from_ipc)."""
        return self._n.arena_ipc_handle(self.handle)

    @classmethod
    def from_ipc(cls, handle: bytes, capacity: int, device: int) -> "Arena":
        """Map another process's device arena (hipIpcOpenMemHandle)."""
        self = cls.__new__(cls)
        self._n = load()
        self.device = device
        self.capacity = capacity
        self.handle = self._n.arena_ipc_open(handle, capacity, device)
        self._closed = False
        return self

    def close(self) -> None:
        if not self._closed:
            self._n.arena_destroy(self.handle)
            self._closed = True

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


class PinnedBuffer:
    """hipHostMalloc'd buffer (plain malloc without a GPU): DMA-fast host
    memory exposed as a writable memoryview.  Used for FUSE request/reply
    buffers so device reads land directly in the bytes handed to writev."""

    def __init__(self, nbytes: int):
        self._n = load()
        self.nbytes = nbytes
        self.id = self._n.pinned_alloc(nbytes)
        self.view = self._n.pinned_view(self.id)
        self.ptr = self._n.pinned_ptr(self.id)
        self._closed = False

    def close(self) -> None:
        if not self._closed:
            self.view = None
            self._n.pinned_free(self.id)
            self._closed = True

    def __del__(self):
        try:
            self.close()
        except Exception:
            pass


def crc32c(buf, init: int = 0) -> int:
    return load().crc32c(buf, init)


def crc32c_combine(crc1: int, crc2: int, len2: int) -> int:
    return load().crc32c_combine(crc1, crc2, len2)


def lz4_compress(buf) -> bytes:
    return load().lz4_compress(buf)


def lz4_decompress(buf) -> bytes:
    return load().lz4_decompress(buf)
