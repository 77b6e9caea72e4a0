"""FsWriter: buffered block writer with star replica fan-out.

Analog of /root/reference/crates/client/curvine-client-core/src/file/
fs_writer.rs + fs_writer_buffer.rs (background flusher) +
block/block_writer.rs:184-210 (each chunk written to ALL replica locations
in parallel — star replication).
"""
from __future__ import annotations

import asyncio
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.client.block_client import make_block_writer
from curvine_amd.client.fs_client import FsClient
from curvine_amd.model import FileStatus, LocatedBlock


class FsWriter:
    def __init__(self, client: FsClient, status: FileStatus):
        self.client = client
        self.status = status
        self.path = status.path
        self.block_size = status.block_size
        self.chunk_size = client.conf.client.write_chunk_size
        self._buf = bytearray()
        self._block: Optional[LocatedBlock] = None
        self._writers: list = []
        self._block_pos = 0
        self._block_lens: list[int] = []
        self._block_addrs: list[tuple] = []   # (block_id, locations, tiers)
        self._rw: dict[int, list] = {}        # rewrite writers per block idx
        self._commits: list[dict] = []   # block locations to report on complete
        self._exclude: set[int] = set()  # workers that failed block opens
        self._abandoned = 0
        self.pos = 0
        self._closed = False
        # end-to-end write integrity: keep a running CRC32C per block and
        # cross-check the worker's publish-time CRC at commit
        self._crc_on = bool(client.conf.client.enable_crc)
        self._blk_crc = 0
        self._crc_valid = True

    async def write(self, data) -> int:
        if self._closed:
            raise err.FsError("writer closed")
        data = memoryview(data)
        total = len(data)
        # zero-copy fast path: with nothing buffered, large inputs go to
        # the block writers directly — no bytearray staging copy, no
        # bytes() slice copy; the event loop thread does no memcpy at all
        # (the copies run on executor threads / in C++ with the GIL
        # released), which is what bounds multi-file ingest throughput
        if not self._buf and total >= self.chunk_size:
            cut = total - (total % self.chunk_size)
            await self._write_view(data[:cut])
            data = data[cut:]
        while len(data) > 0:
            room = self.chunk_size - len(self._buf)
            take = min(room, len(data))
            self._buf.extend(data[:take])
            data = data[take:]
            if len(self._buf) >= self.chunk_size:
                await self._flush_chunk()
        self.pos += total
        return total

    async def _flush_chunk(self) -> None:
        buf = self._buf
        self._buf = bytearray()
        await self._write_view(memoryview(buf))

    async def _write_view(self, buf) -> None:
        off = 0
        while off < len(buf):
            if self._block is None:
                await self._next_block()
            room = self.block_size - self._block_pos
            take = min(room, len(buf) - off)
            chunk = buf[off:off + take]
            # star fan-out: all replicas in parallel
            results = await asyncio.gather(
                *[w.write(chunk) for w in self._writers],
                return_exceptions=True)
            failed = [(self._block.locations[i].worker_id, r)
                      for i, r in enumerate(results)
                      if isinstance(r, BaseException)]
            if failed:
                retriable = all(isinstance(r, (ConnectionError, OSError,
                                               err.ConnectError,
                                               err.RpcTimeout))
                                for _, r in failed)
                if self._block_pos != 0 or not retriable:
                    raise failed[0][1]  # bytes already placed / hard error
                # replica target(s) died before any byte landed: abandon
                # this block (stays 0-length in metadata — readers skip
                # zero-length blocks) and re-place excluding only the
                # failed workers (AddBlockRequest.exclude_workers flow,
                # block_writer.rs replacement-worker analog)
                await self._abandon_block(
                    failed[0][1], [wid for wid, _ in failed])
                continue
            if self._crc_on:
                from curvine_amd import native
                self._blk_crc = native.crc32c(chunk, self._blk_crc)
            self._block_pos += take
            off += take
            if self._block_pos >= self.block_size:
                await self._commit_block()

    async def _abandon_block(self, exc: BaseException,
                             failed_ids: list[int]) -> None:
        for w in self._writers:
            try:
                await w.abort()
            except Exception:  # noqa: BLE001
                pass
        self._exclude.update(failed_ids)
        self._abandoned += 1
        if self._abandoned > 3:
            raise exc
        self._block_lens.append(0)
        self._commits.append({"block_id": self._block.block.block_id,
                              "locations": [], "tiers": []})
        self._block_addrs.append((self._block.block.block_id, [], []))
        self._block = None
        self._writers = []
        self._block_pos = 0

    async def _next_block(self) -> None:
        lb = await self.client.add_block(
            self.path, exclude_workers=sorted(self._exclude) or None)
        self._block = lb
        self._block_pos = 0
        if not lb.locations:
            raise err.NoAvailableWorker(self.path)
        self._writers = [make_block_writer(addr, lb.block.block_id,
                                           self.block_size, tier)
                         for addr, tier in zip(lb.locations, lb.tiers)]

    async def _commit_block(self) -> None:
        tiers = await asyncio.gather(
            *[w.commit(self._block_pos) for w in self._writers])
        if self._crc_on and self._crc_valid:
            for w in self._writers:
                got = getattr(w, "last_crc", None)
                if got is not None and got != self._blk_crc:
                    raise err.ChecksumMismatch(
                        f"block {self._block.block.block_id}: worker crc "
                        f"{got:#010x} != client {self._blk_crc:#010x}")
        self._blk_crc = 0
        self._crc_valid = True
        self._commits.append({
            "block_id": self._block.block.block_id,
            "locations": [a.worker_id for a in self._block.locations],
            "tiers": [t or hint for t, hint in zip(tiers, self._block.tiers)]})
        self._block_lens.append(self._block_pos)
        self._block_addrs.append(
            (self._block.block.block_id, list(self._block.locations),
             [t or hint for t, hint in zip(tiers, self._block.tiers)]))
        self._block = None
        self._writers = []
        self._block_pos = 0

    async def flush(self) -> None:
        if self._buf:
            await self._flush_chunk()

    # ---------------- positional rewrite (writer seek analog) ----------------
    async def pwrite_at(self, off: int, data) -> int:
        """Rewrite already-written bytes in place on every replica
        (fs_writer_base.rs:455-478 seek-write analog; settled blocks are
        reopened positionally).  [off, off+len) must lie within bytes
        this handle has written; appending stays with write()."""
        await self.flush()
        data = memoryview(bytes(data))
        n = len(data)
        if off + n > self.pos:
            raise err.OutOfRange(f"rewrite [{off},{off + n}) past {self.pos}")
        self._crc_valid = False   # running CRC no longer linear
        cur_start = sum(self._block_lens)
        consumed = 0
        while consumed < n:
            o = off + consumed
            if o >= cur_start:                   # current open block
                boff = o - cur_start
                take = min(n - consumed, self._block_pos - boff)
                chunk = bytes(data[consumed:consumed + take])
                await asyncio.gather(
                    *[w.pwrite(boff, chunk) for w in self._writers])
            else:                                # settled block
                import bisect
                starts, s = [], 0
                for ln in self._block_lens:
                    starts.append(s)
                    s += ln
                idx = bisect.bisect_right(starts, o) - 1
                boff = o - starts[idx]
                take = min(n - consumed, self._block_lens[idx] - boff)
                chunk = bytes(data[consumed:consumed + take])
                ws = self._rewrite_writers(idx)
                await asyncio.gather(*[w.pwrite(boff, chunk) for w in ws])
            consumed += take
        return n

    def _rewrite_writers(self, idx: int) -> list:
        ws = self._rw.get(idx)
        if ws is None:
            bid, addrs, tiers = self._block_addrs[idx]
            ws = [make_block_writer(a, bid, 0, t, reopen=True)
                  for a, t in zip(addrs, tiers)]
            self._rw[idx] = ws
        return ws

    async def complete(self) -> FileStatus:
        if self._closed:
            return self.status
        await self.flush()
        if self._block is not None:
            await self._commit_block()
        for ws in self._rw.values():   # close rewrite streams, no finalize
            await asyncio.gather(*[w.commit(None) for w in ws])
        self._rw = {}
        self._closed = True
        length = sum(self._block_lens)
        return await self.client.complete_file(self.path, length,
                                               self._block_lens, self._commits)

    async def abort(self) -> None:
        self._closed = True
        for w in self._writers:
            try:
                await w.abort()
            except Exception:  # noqa: BLE001
                pass
