"""FsReader: positioned reads across blocks with replica fallback.

Analog of /root/reference/crates/client/curvine-client-core/src/file/
fs_reader.rs + block/block_reader.rs:111-230 (local short-circuit first,
remote fallback on failure, failed-worker tracking) and
fs_reader_parallel.rs (sliced parallel reads for large requests);
`ReadDetector` (read_detector.rs:42-134) gates sequential prefetch.
"""
from __future__ import annotations

import asyncio

from curvine_amd import errors as err
from curvine_amd.client.block_client import (BlockReaderHole, BlockReaderLocal,
                                             BlockReaderRemote)
from curvine_amd.client.fs_client import FsClient
from curvine_amd.model import FileBlocks, LocatedBlock


class ReadDetector:
    """Sequential-vs-random pattern detection (read_detector.rs analog)."""

    def __init__(self):
        self.last_end = 0
        self.seq_count = 0
        self.rand_count = 0

    def observe(self, off: int, n: int) -> None:
        if off == self.last_end:
            self.seq_count += 1
        else:
            self.rand_count += 1
        self.last_end = off + n

    @property
    def is_sequential(self) -> bool:
        return self.seq_count >= self.rand_count



def _hole_span(offs, blocks, length, pos) -> int:
    """Length of the zero hole at file offset pos, or 0 when pos falls
    inside a cached block extent.  Metadata length may exceed block
    coverage (extending truncate, sparse tails): the uncovered range
    reads back as zeros, like the reference's hole synthesis
    (block_reader_hole.rs)."""
    import bisect
    idx = bisect.bisect_right(offs, pos) - 1
    nxt = idx + 1
    if idx >= 0:
        lb = blocks[idx]
        if pos < lb.offset + lb.block.length:
            return 0
    end = offs[nxt] if nxt < len(offs) else length
    return max(0, end - pos)


class SyncLocalReader:
    """Synchronous short-circuit reader: all blocks resolved to in-process
    store readers at construction; pread_* are plain function calls safe
    from any OS thread — no event loop in the per-op path.  This is the
    IOPS path (4 KiB random reads must not pay two thread hops + loop
    scheduling per op).  Remote blocks -> construction fails (caller keeps
    the async FsReader)."""

    def __init__(self, file_blocks: FileBlocks, ipc_resolver=None):
        from curvine_amd.worker import registry
        self.fb = file_blocks
        self.length = file_blocks.status.length
        self._offs = [b.offset for b in file_blocks.blocks]
        self._readers = []
        self._native_rid = None
        try:
            for lb in file_blocks.blocks:
                r = None
                for addr in lb.locations:
                    store = registry.lookup(addr.worker_id)
                    if store is not None:
                        r = store.open_reader(lb.block.block_id)
                        break
                if r is None and ipc_resolver is not None:
                    # colocated other-process worker: hipIpc-mapped view
                    r = ipc_resolver(lb)
                if r is None:
                    raise err.BlockNotFound(
                        f"block {lb.block.block_id} has no in-process replica")
                self._readers.append(r)
        except Exception:
            self.close()
            raise
        self._try_register_native()

    def _try_register_native(self) -> None:
        """Pin the extent table in C++ when every block is arena-backed:
        batched preads then resolve+issue+sync GIL-free (reader_register
        in csrc/module.cpp).  The store readers stay open for the whole
        registration (refcounts defer delete/demote), mirroring the
        native FUSE read contract."""
        exts = []
        for lb, r in zip(self.fb.blocks, self._readers):
            meta = r.meta
            arena = getattr(r.layout, "arena", None)
            if meta.get("kind") != "arena" or arena is None:
                return
            exts.append((lb.offset, lb.block.length, arena.handle,
                         meta["offset"]))
        if not exts:
            return
        try:
            from curvine_amd import native
            self._native_rid = native.load().reader_register(
                exts, self.length)
        except Exception:  # noqa: BLE001 — extension absent on this host
            self._native_rid = None

    def pread_into(self, off: int, out, out_off: int, n: int) -> int:
        import bisect
        n = max(0, min(n, self.length - off))
        got = 0
        while got < n:
            pos = off + got
            hole = _hole_span(self._offs, self.fb.blocks, self.length, pos)
            if hole > 0:
                fill = min(hole, n - got)
                out[out_off + got:out_off + got + fill] = b"\x00" * fill
                got += fill
                continue
            idx = bisect.bisect_right(self._offs, pos) - 1
            if idx < 0 or idx >= len(self._readers):
                break
            lb = self.fb.blocks[idx]
            boff = pos - lb.offset
            want = min(n - got, lb.block.length - boff)
            got += self._readers[idx].read_into(boff, out, out_off + got, want)
        return got

    def pread_into_ptr(self, off: int, ptr: int, n: int) -> int:
        """Pinned/device-pointer destination (direct DMA)."""
        import bisect
        import ctypes
        n = max(0, min(n, self.length - off))
        got = 0
        while got < n:
            pos = off + got
            hole = _hole_span(self._offs, self.fb.blocks, self.length, pos)
            if hole > 0:
                fill = min(hole, n - got)
                ctypes.memset(ptr + got, 0, fill)
                got += fill
                continue
            idx = bisect.bisect_right(self._offs, pos) - 1
            if idx < 0 or idx >= len(self._readers):
                break
            lb = self.fb.blocks[idx]
            boff = pos - lb.offset
            want = min(n - got, lb.block.length - boff)
            got += self._readers[idx].read_to_ptr(boff, ptr + got, want, False)
        return got

    def pread(self, off: int, n: int) -> bytes:
        out = bytearray(max(0, min(n, self.length - off)))
        got = self.pread_into(off, out, 0, len(out))
        return bytes(out[:got])

    def pread_batch_ptr(self, file_offs: list[int], n: int, dst_ptr: int,
                        stride: int) -> int:
        """Batched fixed-size reads (fio iodepth analog): read n bytes at
        each file offset into dst_ptr + i*stride.  Arena-resident reads
        issue async on one thread-local stream with ONE sync; anything
        else (file tier, block-spanning) falls back per-read.  Returns the
        number of reads served."""
        import bisect
        import ctypes
        from curvine_amd import native
        if not self.fb.blocks:
            return 0
        if self._native_rid is not None:
            skipped = native.load().reader_pread_batch(
                self._native_rid, file_offs, n, dst_ptr, stride)
            for i in skipped:
                buf = bytearray(n)
                got = self.pread_into(file_offs[i], buf, 0, n)
                ctypes.memmove(dst_ptr + i * stride, bytes(buf[:got]), got)
            return len(file_offs)
        groups: dict[int, tuple[list, list]] = {}  # arena handle -> offs, dsts
        slow: list[tuple[int, int]] = []
        for i, off in enumerate(file_offs):
            idx = max(0, bisect.bisect_right(self._offs, off) - 1)
            lb = self.fb.blocks[idx]
            boff = off - lb.offset
            r = self._readers[idx]
            meta = r.meta
            if meta.get("kind") == "arena" and boff + n <= lb.block.length:
                offs, dsts = groups.setdefault(r.layout.arena.handle, ([], []))
                offs.append(meta["offset"] + boff)
                dsts.append(dst_ptr + i * stride)
            else:
                slow.append((i, off))
        mod = native.load()
        for h, (offs, dsts) in groups.items():
            mod.arena_read_batch(h, offs, dsts, n)
        for i, off in slow:
            buf = bytearray(n)
            got = self.pread_into(off, buf, 0, n)
            ctypes.memmove(dst_ptr + i * stride, bytes(buf[:got]), got)
        return len(file_offs)

    def pread_gather(self, samples: list[tuple[int, int, int]],
                     dst_ptr: int, dst_on_device: bool = False) -> int:
        """Scatter-gather variable-size reads: for each (file_off, n,
        dst_off), land bytes at dst_ptr + dst_off.  Arena-resident extents
        are grouped per arena into ONE gather call (device arenas: the
        on-chip copy_extents_kernel — the HBM-cache -> torch-device-tensor
        training-ingest path with no host hop); everything else falls back
        to pread_into_ptr DMA.  Returns total bytes gathered."""
        groups, slow, total = self.resolve_gather(samples, dst_on_device)
        self.exec_gather(groups, slow, dst_ptr)
        return total

    def resolve_gather(self, samples, dst_on_device: bool):
        """Resolution half of pread_gather, separable so callers with a
        static access pattern (CurvineDeviceLoader) resolve once and
        replay `exec_gather` per epoch.  Valid while this reader is open
        (its store readers pin the blocks)."""
        import bisect
        groups: dict[int, tuple[object, list]] = {}  # handle -> (arena, tri)
        slow: list[tuple[int, int, int]] = []
        total = 0
        for off, n, doff in samples:
            n = max(0, min(n, self.length - off))
            got = 0
            while got < n:
                idx = bisect.bisect_right(self._offs, off + got) - 1
                lb = self.fb.blocks[idx]
                boff = off + got - lb.offset
                want = min(n - got, lb.block.length - boff)
                if want <= 0:
                    break
                r = self._readers[idx]
                meta = r.meta
                arena = getattr(r.layout, "arena", None)
                is_arena = meta.get("kind") == "arena" and arena is not None
                # gather_ptr needs src and dst on the same side (device
                # kernel vs memcpy); cross-side extents use DMA fallback
                same_side = is_arena and \
                    ((arena.device >= 0) == bool(dst_on_device))
                if same_side:
                    _, tri = groups.setdefault(arena.handle, (arena, []))
                    tri.append((meta["offset"] + boff, doff + got, want))
                else:
                    slow.append((off + got, want, doff + got))
                got += want
            total += got
        return list(groups.values()), slow, total

    def exec_gather(self, groups, slow, dst_ptr: int) -> None:
        for arena, tri in groups:
            arena.gather_ptr(tri, dst_ptr)
        for off, want, doff in slow:
            self.pread_into_ptr(off, dst_ptr + doff, want)

    def verify(self) -> list[int]:
        """Recompute every resident block's CRC32C (HBM: device kernel)
        against the worker's publish-time value; returns block ids that
        no longer match (bit rot / unexpected mutation).  Blocks without
        a stored CRC (rewritten in place) are skipped."""
        bad = []
        for lb, r in zip(self.fb.blocks, self._readers):
            stored = r.meta.get("crc32c")
            if stored is None:
                continue
            if r.crc32c(0, lb.block.length) != stored:
                bad.append(lb.block.block_id)
        return bad

    def close(self) -> None:
        if getattr(self, "_native_rid", None) is not None:
            # unregister BEFORE releasing the store readers that pin the
            # registered extents
            try:
                from curvine_amd import native
                native.load().reader_unregister(self._native_rid)
            except Exception:  # noqa: BLE001
                pass
            self._native_rid = None
        for r in getattr(self, "_readers", []):
            try:
                r.close()
            except Exception:  # noqa: BLE001
                pass
        self._readers = []


class FsReader:
    def __init__(self, client: FsClient, file_blocks: FileBlocks):
        self.client = client
        self.fb = file_blocks
        try:
            self._loop = __import__("asyncio").get_running_loop()
        except RuntimeError:
            self._loop = None
        self.status = file_blocks.status
        self.length = self.status.length
        self.pos = 0
        self.detector = ReadDetector()
        self._readers: dict[int, object] = {}   # block index -> reader
        self.failed_workers: set[int] = set()
        self.chunk_size = client.conf.client.read_chunk_size
        self.parallel = max(1, client.conf.client.read_parallel)
        self.slice_size = client.conf.client.read_slice_size

    # ---------------- block reader selection ----------------
    def _block_at(self, off: int) -> tuple[int, LocatedBlock, int]:
        """(index, located block, offset within block).  Blocks can be
        shorter than block_size mid-file (appends), so search by offset."""
        import bisect
        offs = getattr(self, "_block_offs", None)
        if offs is None or len(offs) != len(self.fb.blocks):
            offs = [b.offset for b in self.fb.blocks]
            self._block_offs = offs
        idx = bisect.bisect_right(offs, off) - 1
        if idx < 0 or idx >= len(self.fb.blocks):
            raise err.OutOfRange(f"offset {off} beyond {self.length}")
        lb = self.fb.blocks[idx]
        return idx, lb, off - lb.offset

    async def _reader_for(self, idx: int):
        r = self._readers.get(idx)
        if r is not None:
            return r
        lb = self.fb.blocks[idx]
        r = await self._open_block_reader(lb)
        self._readers[idx] = r
        return r

    async def _open_block_reader(self, lb: LocatedBlock):
        if not lb.locations:
            # no cached replica: sparse hole (or caller falls back to UFS)
            return BlockReaderHole(lb.block.length)
        from curvine_amd.worker import registry
        candidates = [a for a in lb.locations
                      if a.worker_id not in self.failed_workers] or lb.locations
        # in-process short-circuit first
        if self.client.conf.client.short_circuit:
            for addr in candidates:
                store = registry.lookup(addr.worker_id)
                if store is not None:
                    try:
                        return BlockReaderLocal(store, lb.block.block_id)
                    except Exception:  # noqa: BLE001
                        self.failed_workers.add(addr.worker_id)
        # cross-process HBM short-circuit: a colocated worker in another
        # process discloses its arena via hipIpc — direct DMA reads here
        if self.client.conf.client.short_circuit:
            from curvine_amd.client.block_client import (AsyncIpcReader,
                                                         open_ipc_reader)
            for addr in candidates:
                ipc = await open_ipc_reader(self.client, addr,
                                            lb.block.block_id)
                if ipc is not None:
                    return AsyncIpcReader(ipc)
        last: Exception = err.BlockNotFound(str(lb.block.block_id))
        for addr in candidates:
            try:
                r = BlockReaderRemote(addr, lb.block.block_id)
                return r
            except Exception as e:  # noqa: BLE001
                self.failed_workers.add(addr.worker_id)
                last = e
        raise last

    # ---------------- reads ----------------
    async def pread(self, off: int, n: int) -> bytes:
        n = max(0, min(n, self.length - off))
        if n == 0:
            return b""
        out = bytearray(n)
        got = await self.pread_into(off, out, 0, n)
        return bytes(out[:got])

    async def pread_into(self, off: int, out, out_off: int, n: int) -> int:
        n = max(0, min(n, self.length - off))
        if n == 0:
            return 0
        self.detector.observe(off, n)
        if n >= self.slice_size and self.parallel > 1:
            return await self._pread_parallel(off, out, out_off, n)
        got = 0
        while got < n:
            pos = off + got
            hole = _hole_span(self._block_offsets(), self.fb.blocks,
                              self.length, pos)
            if hole > 0:
                fill = min(hole, n - got)
                out[out_off + got:out_off + got + fill] = b"\x00" * fill
                got += fill
                continue
            idx, lb, boff = self._block_at(pos)
            want = min(n - got, lb.block.length - boff)
            if want <= 0:
                break
            r = await self._reader_for(idx)
            rn = await self._read_with_fallback(idx, r, boff, out,
                                                out_off + got, want)
            if rn <= 0:
                break
            got += rn
        return got

    def _block_offsets(self) -> list:
        offs = getattr(self, "_block_offs", None)
        if offs is None or len(offs) != len(self.fb.blocks):
            offs = [b.offset for b in self.fb.blocks]
            self._block_offs = offs
        return offs

    async def _read_with_fallback(self, idx, reader, boff, out, out_off, want):
        try:
            return await reader.read_into(boff, out, out_off, want)
        except Exception:  # noqa: BLE001 — replica failed mid-read
            self._readers.pop(idx, None)
            lb = self.fb.blocks[idx]
            r2 = await self._open_block_reader(lb)
            self._readers[idx] = r2
            return await r2.read_into(boff, out, out_off, want)

    async def _pread_parallel(self, off, out, out_off, n) -> int:
        """Slice fan-out (fs_reader_parallel.rs:94-131 analog)."""
        k = min(self.parallel, (n + self.slice_size - 1) // self.slice_size)
        per = (n + k - 1) // k
        tasks = []
        for i in range(k):
            so = off + i * per
            sn = min(per, off + n - so)
            if sn <= 0:
                break
            tasks.append(self._pread_slice(so, out, out_off + i * per, sn))
        results = await asyncio.gather(*tasks)
        return sum(results)

    async def _pread_slice(self, off, out, out_off, n) -> int:
        got = 0
        while got < n:
            pos = off + got
            hole = _hole_span(self._block_offsets(), self.fb.blocks,
                              self.length, pos)
            if hole > 0:
                fill = min(hole, n - got)
                out[out_off + got:out_off + got + fill] = b"\x00" * fill
                got += fill
                continue
            idx, lb, boff = self._block_at(pos)
            want = min(n - got, lb.block.length - boff)
            if want <= 0:
                break
            r = await self._reader_for(idx)
            rn = await self._read_with_fallback(idx, r, boff, out,
                                                out_off + got, want)
            if rn <= 0:
                break
            got += rn
        return got

    async def read(self, n: int = -1) -> bytes:
        if n < 0:
            n = self.length - self.pos
        data = await self.pread(self.pos, n)
        self.pos += len(data)
        return data

    def seek(self, pos: int) -> None:
        self.pos = pos

    async def pread_to_device(self, off: int, dst_ptr: int, n: int) -> int:
        """Read file bytes directly into consumer GPU memory
        (hipMemcpyDtoD from the HBM arena when local, staged otherwise)."""
        n = max(0, min(n, self.length - off))
        got = 0
        while got < n:
            idx, lb, boff = self._block_at(off + got)
            want = min(n - got, lb.block.length - boff)
            if want <= 0:
                break
            r = await self._reader_for(idx)
            if isinstance(r, BlockReaderLocal):
                rn = await r.read_to_device(boff, dst_ptr + got, want)
            else:
                # remote block: fetch to pinned host memory, then H2D
                from curvine_amd import native
                pbuf = native.PinnedBuffer(want)
                try:
                    rn = await r.read_into(boff, pbuf.view, 0, want)
                    if rn > 0:
                        loop = asyncio.get_event_loop()
                        await loop.run_in_executor(
                            None, native.load().memcpy_h2d,
                            dst_ptr + got, pbuf.ptr, rn)
                finally:
                    pbuf.close()
            got += rn
        return got

    def to_sync(self) -> SyncLocalReader:
        """Short-circuit sync view (raises if any block lacks a local
        in-process OR hipIpc-mapped replica)."""
        return SyncLocalReader(self.fb, ipc_resolver=self._ipc_resolver)

    def _ipc_resolver(self, lb):
        """Sync bridge for SyncLocalReader: map a colocated
        other-process HBM block via the client's event loop."""
        import asyncio as _a
        if not self.client.conf.client.short_circuit or self._loop is None:
            return None
        try:
            if _a.get_running_loop() is self._loop:
                return None   # would deadlock; caller stays async
        except RuntimeError:
            pass
        from curvine_amd.client.block_client import open_ipc_reader
        for addr in lb.locations:
            try:
                fut = _a.run_coroutine_threadsafe(
                    open_ipc_reader(self.client, addr, lb.block.block_id),
                    self._loop)
                ipc = fut.result(30)
            except Exception:  # noqa: BLE001
                continue
            if ipc is not None:
                return ipc
        return None

    def close(self) -> None:
        for r in self._readers.values():
            try:
                r.close()
            except Exception:  # noqa: BLE001
                pass
        self._readers.clear()
