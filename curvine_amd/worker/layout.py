"""Block layouts: where block bytes live.

Analog of the reference's `BlockLayout` trait + implementations
(/root/reference/crates/adapters/curvine-storage-local/src/layout/mod.rs:
36-60, file_layout.rs:29-212 one-file-per-block, bdev_layout.rs:30-111
raw-extent): here

* `ArenaLayout` — blocks are extents in a `native.Arena`: HBM tier
  (device >= 0, the MI355X hot tier) or host-memory MEM tier (device -1).
* `FileLayout` — one file per block (SSD/HDD/NVMe tiers), with
  writing/finalized state prefixes like the reference.
"""
from __future__ import annotations

import os

from curvine_amd import errors as err
from curvine_amd.conf import DataDir, TIER_HBM, TIER_MEM
from curvine_amd.native import Arena


class BlockWriter:
    """Append-only writer for one block."""

    def __init__(self, layout: "BlockLayout", block_id: int, meta: dict):
        self.layout = layout
        self.block_id = block_id
        self.meta = meta
        self.pos = 0

    def write(self, data, n: int | None = None) -> int:
        n = len(data) if n is None else n
        self.layout._write_at(self.meta, self.pos, data, n)
        self.pos += n
        return n

    def write_from_ptr(self, ptr: int, n: int, device: bool) -> int:
        self.layout._write_at_ptr(self.meta, self.pos, ptr, n, device)
        self.pos += n
        return n

    def pwrite(self, off: int, data, n: int | None = None) -> int:
        """Positional write within the reserved allocation (random-write
        support, fs_writer_base.rs:455-478 seek analog); `pos` tracks the
        high watermark used as the default finalize length."""
        n = len(data) if n is None else n
        self.layout._write_at(self.meta, off, data, n)
        self.pos = max(self.pos, off + n)
        return n


class BlockReader:
    def __init__(self, layout: "BlockLayout", block_id: int, meta: dict):
        self.layout = layout
        self.block_id = block_id
        self.meta = meta
        self.length = meta["length"]

    def read(self, off: int, n: int) -> bytes:
        n = min(n, self.length - off)
        if n <= 0:
            return b""
        return self.layout._read_at(self.meta, off, n)

    def read_into(self, off: int, out, out_off: int, n: int) -> int:
        n = min(n, self.length - off)
        if n <= 0:
            return 0
        self.layout._read_into(self.meta, off, out, out_off, n)
        return n

    def read_to_ptr(self, off: int, dst_ptr: int, n: int, device: bool) -> int:
        n = min(n, self.length - off)
        if n > 0:
            self.layout._read_to_ptr(self.meta, off, dst_ptr, n, device)
        return n

    def crc32c(self, off: int, n: int) -> int:
        return self.layout._crc(self.meta, off, min(n, self.length - off))

    def close(self) -> None:
        pass


class BlockLayout:
    tier: str = "?"

    def __init__(self, data_dir: DataDir, dir_id: int):
        self.conf = data_dir
        self.dir_id = dir_id
        self.capacity = data_dir.capacity

    # capacity
    @property
    def used(self) -> int:
        raise NotImplementedError

    @property
    def available(self) -> int:
        return max(0, self.capacity - self.used)

    # lifecycle
    def allocate(self, block_id: int, reserve: int) -> dict:
        """Reserve space; returns layout meta for the block."""
        raise NotImplementedError

    def finalize(self, meta: dict, length: int) -> None:
        raise NotImplementedError

    def deallocate(self, meta: dict) -> None:
        raise NotImplementedError

    def scan(self) -> list[dict]:
        """Startup scan: recover finalized blocks (vfs_dataset.rs:197-213)."""
        return []

    # io
    def _write_at(self, meta: dict, off: int, data, n: int) -> None:
        raise NotImplementedError

    def _read_at(self, meta: dict, off: int, n: int) -> bytes:
        raise NotImplementedError

    def _read_into(self, meta: dict, off: int, out, out_off: int, n: int) -> None:
        raise NotImplementedError

    def _read_to_ptr(self, meta: dict, off: int, ptr: int, n: int, device: bool) -> None:
        raise NotImplementedError

    def _write_at_ptr(self, meta: dict, off: int, ptr: int, n: int, device: bool) -> None:
        raise NotImplementedError

    def _crc(self, meta: dict, off: int, n: int) -> int:
        raise NotImplementedError

    def local_info(self, meta: dict) -> dict:
        """Short-circuit disclosure (block_store.rs:253-271 analog)."""
        raise NotImplementedError

    def close(self) -> None:
        pass


class ArenaLayout(BlockLayout):
    """HBM (device) or MEM (host) arena-backed blocks."""

    def __init__(self, data_dir: DataDir, dir_id: int,
                 staging_bytes: int = 4 << 20, staging_count: int = 8):
        super().__init__(data_dir, dir_id)
        self.tier = data_dir.tier
        device = data_dir.device_id if data_dir.tier == TIER_HBM else -1
        from curvine_amd.worker.arena_alloc import ArenaAllocator
        self.arena = Arena(device, data_dir.capacity, staging_bytes,
                           staging_count,
                           host_pinned=(data_dir.tier == TIER_MEM and device < 0))
        self.allocator = ArenaAllocator(data_dir.capacity)

    @property
    def used(self) -> int:
        return self.allocator.used

    def allocate(self, block_id: int, reserve: int) -> dict:
        off = self.allocator.alloc(reserve)
        return {"kind": "arena", "tier": self.tier, "dir_id": self.dir_id,
                "offset": off, "reserved": reserve, "length": 0,
                "device": self.arena.device}

    def finalize(self, meta: dict, length: int) -> None:
        meta["length"] = length
        self.allocator.shrink(meta["offset"], length)

    def deallocate(self, meta: dict) -> None:
        self.allocator.release(meta["offset"])

    def _write_at(self, meta, off, data, n):
        self.arena.write(meta["offset"] + off, data, 0, n)

    def _write_at_ptr(self, meta, off, ptr, n, device):
        self.arena.write_from_ptr(meta["offset"] + off, ptr, n, device)

    def _read_at(self, meta, off, n):
        return self.arena.read_bytes(meta["offset"] + off, n)

    def _read_into(self, meta, off, out, out_off, n):
        self.arena.read(meta["offset"] + off, out, out_off, n)

    def _read_to_ptr(self, meta, off, ptr, n, device):
        self.arena.read_to_ptr(meta["offset"] + off, ptr, n, device)

    def _crc(self, meta, off, n):
        return self.arena.crc32c(meta["offset"] + off, n)

    def local_info(self, meta: dict) -> dict:
        info = {"kind": "arena", "tier": self.tier,
                "arena_handle": self.arena.handle,
                "device": self.arena.device,
                "offset": meta["offset"], "length": meta["length"]}
        if self.arena.device >= 0:
            # cross-process short-circuit: a colocated client opens the
            # hipIpc handle and DMA-reads the extent in its own process
            if not hasattr(self, "_ipc"):
                try:
                    self._ipc = self.arena.ipc_handle()
                    self._ipc_err = None
                except Exception as e:  # noqa: BLE001 — pool without IPC
                    self._ipc = None
                    self._ipc_err = f"{type(e).__name__}: {e}"
            if self._ipc:
                info["ipc"] = self._ipc
                info["cap"] = self.arena.capacity
            elif self._ipc_err:
                info["ipc_error"] = self._ipc_err
        return info

    def close(self) -> None:
        self.arena.close()


class FileLayout(BlockLayout):
    """One file per block: <dir>/<state>/<id % 256>/<id>."""

    WRITING = "writing"
    FINAL = "final"

    ALIGN = 4096   # O_DIRECT sector/buffer alignment

    def __init__(self, data_dir: DataDir, dir_id: int):
        super().__init__(data_dir, dir_id)
        self.tier = data_dir.tier
        self.root = data_dir.path
        # O_DIRECT reads (page-cache bypass on NVMe; writes stay buffered
        # with fsync at finalize + DONTNEED so the cache stays clean —
        # the read path is what a cache serves)
        self.direct = data_dir.o_direct
        os.makedirs(os.path.join(self.root, self.WRITING), exist_ok=True)
        os.makedirs(os.path.join(self.root, self.FINAL), exist_ok=True)
        if self.direct:
            self.direct = self._probe_direct()
        import threading
        self._used = 0
        self._used_lock = threading.Lock()
        self._fds: dict[int, object] = {}

    def _probe_direct(self) -> bool:
        """tmpfs and some overlays reject O_DIRECT: probe once and fall
        back to buffered reads with a warning instead of failing every
        read."""
        probe = os.path.join(self.root, ".direct_probe")
        try:
            with open(probe, "wb") as f:
                f.write(b"\0" * self.ALIGN)
            fd = os.open(probe, os.O_RDONLY | os.O_DIRECT)
            os.close(fd)
            return True
        except OSError as e:
            import logging
            logging.getLogger("curvine.layout").warning(
                "dir %s: O_DIRECT unsupported (%s); buffered reads",
                self.root, e)
            return False
        finally:
            try:
                os.remove(probe)
            except OSError:
                pass

    def _pread_direct(self, path: str, off: int, n: int) -> bytes:
        """Aligned O_DIRECT pread covering [off, off+n) (bounce through
        an mmap page-aligned buffer; short tail reads clamp at EOF)."""
        import mmap
        a = self.ALIGN
        lo = off - (off % a)
        span = off + n - lo
        span = span + (-span % a)
        fd = os.open(path, os.O_RDONLY | os.O_DIRECT)
        try:
            buf = mmap.mmap(-1, span)
            got = os.preadv(fd, [buf], lo)
            return buf[off - lo:min(off - lo + n, got)]
        finally:
            os.close(fd)

    def _path(self, block_id: int, state: str) -> str:
        sub = os.path.join(self.root, state, f"{block_id % 256:02x}")
        os.makedirs(sub, exist_ok=True)
        return os.path.join(sub, str(block_id))

    @property
    def used(self) -> int:
        return self._used

    def allocate(self, block_id: int, reserve: int) -> dict:
        # reserve against capacity FIRST (dir_state.rs:20 analog): a full
        # SSD/HDD dir must raise so BlockStore.create_writer's tier
        # fall-through can move to the next dir instead of overcommitting
        # the disk
        with self._used_lock:
            if self._used + reserve > self.capacity:
                raise err.CapacityExceeded(
                    f"dir {self.root}: used {self._used} + reserve "
                    f"{reserve} > capacity {self.capacity}")
            self._used += reserve
        try:
            path = self._path(block_id, self.WRITING)
            f = open(path, "wb")
        except OSError:
            with self._used_lock:
                self._used -= reserve
            raise
        return {"kind": "file", "tier": self.tier, "dir_id": self.dir_id,
                "path": path, "block_id": block_id, "reserved": reserve,
                "length": 0, "_f": f}

    fsync_on_finalize = False   # set from WorkerConf by BlockStore

    def finalize(self, meta: dict, length: int) -> None:
        f = meta.pop("_f", None)
        if f:
            f.flush()
            if self.fsync_on_finalize:
                os.fsync(f.fileno())
            if self.direct:
                # reads bypass the cache: drop the write-side pages now
                try:
                    os.posix_fadvise(f.fileno(), 0, 0,
                                     os.POSIX_FADV_DONTNEED)
                except OSError:
                    pass
            f.close()
        final = self._path(meta["block_id"], self.FINAL)
        os.replace(meta["path"], final)
        meta["path"] = final
        meta["length"] = length
        with self._used_lock:
            self._used -= meta["reserved"] - length
        meta["reserved"] = length

    def deallocate(self, meta: dict) -> None:
        f = meta.pop("_f", None)
        if f:
            try:
                f.close()
            except Exception:  # noqa: BLE001
                pass
        try:
            os.remove(meta["path"])
        except FileNotFoundError:
            pass
        with self._used_lock:
            self._used -= meta["reserved"]

    def scan(self) -> list[dict]:
        out = []
        final_root = os.path.join(self.root, self.FINAL)
        for dirpath, _, files in os.walk(final_root):
            for name in files:
                if not name.isdigit():
                    continue
                path = os.path.join(dirpath, name)
                ln = os.path.getsize(path)
                out.append({"kind": "file", "tier": self.tier,
                            "dir_id": self.dir_id, "path": path,
                            "block_id": int(name), "reserved": ln,
                            "length": ln})
                self._used += ln
        # writing/ leftovers are partial: discard (crash recovery)
        writing_root = os.path.join(self.root, self.WRITING)
        for dirpath, _, files in os.walk(writing_root):
            for name in files:
                try:
                    os.remove(os.path.join(dirpath, name))
                except OSError:
                    pass
        return out

    def _write_at(self, meta, off, data, n):
        f = meta.get("_f")
        if f is None:
            raise err.FsError("block not open for write")
        f.seek(off)
        mv = memoryview(data)[:n] if n != len(data) else data
        f.write(mv)

    def _write_at_ptr(self, meta, off, ptr, n, device):
        raise err.Unsupported("ptr write to file layout")

    def _read_at(self, meta, off, n):
        if self.direct:
            return self._pread_direct(meta["path"], off, n)
        with open(meta["path"], "rb") as f:
            f.seek(off)
            return f.read(n)

    def _read_into(self, meta, off, out, out_off, n):
        if self.direct:
            data = self._pread_direct(meta["path"], off, n)
            out[out_off:out_off + len(data)] = data
            return
        with open(meta["path"], "rb") as f:
            f.seek(off)
            mv = memoryview(out)[out_off:out_off + n]
            f.readinto(mv)

    def _read_to_ptr(self, meta, off, ptr, n, device):
        raise err.Unsupported("ptr read from file layout")

    def _crc(self, meta, off, n):
        from curvine_amd.native import crc32c, crc32c_combine
        crc = 0
        first = True
        with open(meta["path"], "rb") as f:
            f.seek(off)
            remain = n
            while remain > 0:
                chunk = f.read(min(remain, 8 << 20))
                if not chunk:
                    break
                c = crc32c(chunk)
                crc = c if first else crc32c_combine(crc, c, len(chunk))
                first = False
                remain -= len(chunk)
        return crc

    def local_info(self, meta: dict) -> dict:
        return {"kind": "file", "tier": self.tier, "path": meta["path"],
                "length": meta["length"]}


def make_layout(data_dir: DataDir, dir_id: int, staging_bytes: int = 4 << 20,
                staging_count: int = 8) -> BlockLayout:
    if data_dir.tier in (TIER_HBM, TIER_MEM):
        return ArenaLayout(data_dir, dir_id, staging_bytes, staging_count)
    return FileLayout(data_dir, dir_id)
