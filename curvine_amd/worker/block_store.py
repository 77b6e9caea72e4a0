"""Worker block store: multi-tier facade with reservation/rollback.

Analog of the reference's `BlockStore` + `VfsDataset`
(/root/reference/curvine-worker/src/worker/block/block_store.rs:27-330:
open/finalize with reservation-prepare-publish and rollback :128-193,
generation-safe readers :234-248, short-circuit disclosure :253-271;
crates/adapters/curvine-storage-local/src/vfs_dataset.rs:31-213 multi-dir
multi-tier dataset with startup scan).

Tier order is hottest-first: HBM (GPU arenas) > MEM (host) > SSD > HDD;
allocation falls through to colder tiers on capacity pressure.  Metadata
mutations hold a lock; byte movement happens outside it.
"""
from __future__ import annotations

import logging
import threading

from curvine_amd import errors as err
from curvine_amd.conf import TIER_ORDER, WorkerConf
from curvine_amd.model import BlockState, StorageInfo
from curvine_amd.worker.layout import (BlockLayout, BlockReader, BlockWriter,
                                       make_layout)

log = logging.getLogger("curvine.blockstore")


class _Block:
    __slots__ = ("block_id", "state", "meta", "layout", "generation",
                 "readers", "pending_delete", "last_access")

    def __init__(self, block_id: int, layout: BlockLayout, meta: dict):
        self.block_id = block_id
        self.state = BlockState.WRITING
        self.meta = meta
        self.layout = layout
        self.generation = 0
        self.readers = 0
        self.pending_delete = False
        self.last_access = 0.0


class BlockStore:
    def __init__(self, conf: WorkerConf):
        self.conf = conf
        self.layouts: list[BlockLayout] = []
        for i, dd in enumerate(conf.parsed_dirs()):
            layout = make_layout(dd, i, conf.staging_buf_bytes,
                                 conf.staging_buf_count)
            layout.fsync_on_finalize = conf.fsync_on_finalize
            self.layouts.append(layout)
        # hottest-first iteration order
        self.layouts.sort(key=lambda l: TIER_ORDER.get(l.tier, 9))
        self.blocks: dict[int, _Block] = {}
        self.lock = threading.Lock()
        # incremental heartbeat deltas
        self._added: list[dict] = []
        self._removed: list[int] = []
        # native data plane observer (worker/native_data.py): finalized
        # blocks are published to the C++ read registry; deletes with
        # in-flight native readers defer the layout free (reap_deferred)
        self.data_plane = None
        self._deferred_frees: list[tuple[int, object, dict]] = []
        self.scan()

    # ---------------- startup ----------------
    def scan(self) -> None:
        for layout in self.layouts:
            for meta in layout.scan():
                b = _Block(meta["block_id"], layout, meta)
                b.state = BlockState.FINALIZED
                self.blocks[b.block_id] = b
                self._added.append({"block_id": b.block_id, "tier": layout.tier})
        if self.blocks:
            log.info("block store recovered %d blocks", len(self.blocks))

    # ---------------- write path ----------------
    def create_writer(self, block_id: int, reserve: int,
                      tier_hint: str = "") -> BlockWriter:
        from curvine_amd.fault import fault_point
        fault_point("worker.block.create")
        with self.lock:
            existing = self.blocks.get(block_id)
            if existing is not None:
                if existing.state == BlockState.WRITING:
                    raise err.BlockInWriting(str(block_id))
                raise err.FileAlreadyExists(f"block {block_id}")
        # allocation outside the map lock (layouts have their own state)
        start = 0
        if tier_hint in TIER_ORDER:
            start = TIER_ORDER[tier_hint]
        last_exc: Exception = err.CapacityExceeded("no data dirs")
        for layout in self.layouts:
            if TIER_ORDER.get(layout.tier, 9) < start:
                continue
            try:
                meta = layout.allocate(block_id, reserve)
            except err.CapacityExceeded as e:
                # allocation pressure: reclaim drained deferred deletes
                # now instead of waiting for the heartbeat tick, then
                # retry this tier once
                if self.reap_deferred() == 0:
                    last_exc = e
                    continue
                try:
                    meta = layout.allocate(block_id, reserve)
                except err.CapacityExceeded as e2:
                    last_exc = e2
                    continue
            b = _Block(block_id, layout, meta)
            with self.lock:
                if block_id in self.blocks:   # raced: rollback
                    layout.deallocate(meta)
                    raise err.BlockInWriting(str(block_id))
                self.blocks[block_id] = b
            return BlockWriter(layout, block_id, meta)
        raise last_exc

    def reopen_writer(self, block_id: int) -> BlockWriter:
        """Positional-rewrite access to an existing block (random writes:
        the reference reopens a block writer at the seek position,
        fs_writer_base.rs:466-473).  FINALIZED blocks may be rewritten IN
        PLACE within their length (concurrent readers see unspecified
        interleaving, as POSIX allows); growth requires the append path."""
        with self.lock:
            b = self.blocks.get(block_id)
            if b is None or b.pending_delete:
                raise err.BlockNotFound(str(block_id))
        w = BlockWriter(b.layout, block_id, b.meta)
        if b.state == BlockState.FINALIZED:
            w.pos = b.meta.get("length", 0)
            b.meta.pop("crc32c", None)   # in-place rewrite invalidates it
        return w

    def finalize(self, block_id: int, length: int) -> str:
        """Publish a written block; returns its tier."""
        from curvine_amd.fault import fault_point
        fault_point("worker.block.finalize")
        with self.lock:
            b = self.blocks.get(block_id)
            if b is None:
                raise err.BlockNotFound(str(block_id))
            if b.state == BlockState.FINALIZED:
                return b.layout.tier
        b.layout.finalize(b.meta, length)
        # block checksum at publish time (HBM: the CRC32C kernel at
        # ~2.4 TB/s; host tiers: SSE4.2) — end-to-end write integrity
        # (clients with enable_crc cross-check this against their own
        # running CRC) and later verify() support
        try:
            b.meta["crc32c"] = b.layout._crc(b.meta, 0, length) if length else 0
        except Exception:  # noqa: BLE001 — integrity is best-effort here
            b.meta.pop("crc32c", None)
        with self.lock:
            b.state = BlockState.FINALIZED
            b.generation += 1
            self._added.append({"block_id": block_id, "tier": b.layout.tier})
        if self.data_plane is not None:
            self.data_plane.publish(block_id, b.layout, b.meta)
        return b.layout.tier

    def block_crc(self, block_id: int):
        """Stored publish-time CRC32C, or None (unpublished / rewritten)."""
        with self.lock:
            b = self.blocks.get(block_id)
            return None if b is None else b.meta.get("crc32c")

    def abort(self, block_id: int) -> None:
        with self.lock:
            b = self.blocks.pop(block_id, None)
        if b is not None and b.state == BlockState.WRITING:
            b.layout.deallocate(b.meta)
        elif b is not None:
            # was finalized; put it back
            with self.lock:
                self.blocks[block_id] = b

    # ---------------- read path ----------------
    def open_reader(self, block_id: int) -> BlockReader:
        with self.lock:
            b = self.blocks.get(block_id)
            if b is None or b.pending_delete:
                raise err.BlockNotFound(str(block_id))
            if b.state != BlockState.FINALIZED:
                raise err.BlockInWriting(str(block_id))
            b.readers += 1
            import time as _time
            b.last_access = _time.monotonic()
            meta = b.meta
        reader = BlockReader(b.layout, block_id, meta)
        store = self

        def close():
            with store.lock:
                b.readers -= 1
                if b.pending_delete and b.readers == 0:
                    store._do_delete(b)
        reader.close = close  # type: ignore[method-assign]
        return reader

    def local_info(self, block_id: int) -> dict:
        with self.lock:
            b = self.blocks.get(block_id)
            if b is None or b.state != BlockState.FINALIZED or b.pending_delete:
                raise err.BlockNotFound(str(block_id))
            return b.layout.local_info(b.meta)

    # ---------------- delete ----------------
    def delete(self, block_id: int) -> None:
        with self.lock:
            b = self.blocks.get(block_id)
            if b is None:
                return
            if b.readers > 0:
                b.pending_delete = True
                return
            self._do_delete(b)

    def _do_delete(self, b: _Block) -> None:
        # caller holds self.lock
        self.blocks.pop(b.block_id, None)
        self._removed.append(b.block_id)
        if self.data_plane is not None and \
                self.data_plane.drop(b.block_id) > 0:
            # native readers still streaming this extent: defer the free
            # until they finish (reap_deferred polls refs)
            self._deferred_frees.append((b.block_id, b.layout, b.meta))
            return
        try:
            b.layout.deallocate(b.meta)
        except Exception as e:  # noqa: BLE001
            log.warning("deallocate block %d: %s", b.block_id, e)

    def reap_deferred(self) -> int:
        """Free deferred deletes whose native readers have drained
        (called from the worker heartbeat loop)."""
        if self.data_plane is None or not self._deferred_frees:
            return 0
        freed = 0
        with self.lock:
            pending, self._deferred_frees = self._deferred_frees, []
        for bid, layout, meta in pending:
            if self.data_plane.refs(bid) > 0:
                with self.lock:
                    self._deferred_frees.append((bid, layout, meta))
                continue
            try:
                layout.deallocate(meta)
                freed += 1
            except Exception as e:  # noqa: BLE001
                log.warning("deferred deallocate block %d: %s", bid, e)
        return freed

    # ---------------- reporting ----------------
    def take_deltas(self) -> tuple[list[dict], list[int]]:
        with self.lock:
            added, self._added = self._added, []
            removed, self._removed = self._removed, []
        return added, removed

    def full_report(self) -> list[dict]:
        with self.lock:
            return [{"block_id": b.block_id, "tier": b.layout.tier,
                     "length": b.meta.get("length", 0)}
                    for b in self.blocks.values()
                    if b.state == BlockState.FINALIZED and not b.pending_delete]

    def storages(self) -> list[StorageInfo]:
        with self.lock:
            counts: dict[int, int] = {}
            for b in self.blocks.values():
                counts[b.layout.dir_id] = counts.get(b.layout.dir_id, 0) + 1
        return [StorageInfo(tier=l.tier, dir_id=l.dir_id, capacity=l.capacity,
                            used=l.used, block_num=counts.get(l.dir_id, 0))
                for l in self.layouts]

    def block_count(self) -> int:
        with self.lock:
            return len(self.blocks)

    # ---------------- tier demotion ----------------
    def demote_coldest(self, high_watermark: float = 0.90,
                       low_watermark: float = 0.75) -> int:
        """Move least-recently-read blocks from a pressured tier to the
        next colder one (HBM -> MEM/SSD demotion; the MI355X analog of the
        reference's multi-tier rebalancing).  Finalized blocks are
        immutable, so the copy runs without blocking readers; the swap is
        skipped if a reader appeared meanwhile.  Returns blocks moved."""
        moved = 0
        for i, src in enumerate(self.layouts[:-1]):
            if src.capacity <= 0 or src.used < src.capacity * high_watermark:
                continue
            target_used = src.capacity * low_watermark
            with self.lock:
                cands = sorted(
                    (b for b in self.blocks.values()
                     if b.layout is src and b.state == BlockState.FINALIZED
                     and not b.pending_delete),
                    key=lambda b: b.last_access)
            for b in cands:
                if src.used <= target_used:
                    break
                if self._demote_one(b, self.layouts[i + 1:]):
                    moved += 1
        return moved

    def _demote_one(self, b: _Block, colder: list[BlockLayout]) -> bool:
        length = b.meta.get("length", 0)
        for dst in colder:
            if dst.available < length:
                continue
            try:
                new_meta = dst.allocate(b.block_id, length)
            except err.CapacityExceeded:
                continue
            try:
                self._copy_block(b.layout, b.meta, dst, new_meta, length)
                dst.finalize(new_meta, length)
            except Exception as e:  # noqa: BLE001
                log.warning("demotion copy of block %d failed: %s",
                            b.block_id, e)
                dst.deallocate(new_meta)
                return False
            with self.lock:
                if b.readers > 0 or b.pending_delete or \
                        self.blocks.get(b.block_id) is not b:
                    swap = False
                else:
                    old_layout, old_meta = b.layout, b.meta
                    b.layout, b.meta = dst, new_meta
                    self._added.append({"block_id": b.block_id,
                                        "tier": dst.tier})
                    swap = True
            if swap:
                old_layout.deallocate(old_meta)
                log.info("demoted block %d %s -> %s (%d bytes)",
                         b.block_id, old_layout.tier, dst.tier, length)
                return True
            dst.deallocate(new_meta)
            return False
        return False

    @staticmethod
    def _copy_block(src_layout, src_meta, dst_layout, dst_meta,
                    length: int, chunk: int = 16 << 20) -> None:
        from curvine_amd.worker.layout import ArenaLayout
        if isinstance(src_layout, ArenaLayout) and \
                isinstance(dst_layout, ArenaLayout):
            # device/host arena direct copy (D2D 5 TB/s on-chip, or DMA)
            src_layout.arena.copy_to(dst_layout.arena, dst_meta["offset"],
                                     src_meta["offset"], length)
            return
        pos = 0
        while pos < length:
            n = min(chunk, length - pos)
            data = src_layout._read_at(src_meta, pos, n)
            dst_layout._write_at(dst_meta, pos, data, len(data))
            pos += n

    def close(self) -> None:
        for l in self.layouts:
            l.close()
