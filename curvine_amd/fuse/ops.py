"""FUSE operations over the curvine client.

Analog of the reference's `CurvineFileSystem` (fuse side,
/root/reference/curvine-fuse/src/fs/curvine_file_system.rs, 3.7k lines)
with its dcache (fs/dcache/dir_tree.rs:44-93 ino<->path map), handle state
(fs/state/), FuseReader/FuseWriter stream bridges (fs/fuse_reader.rs:37-122,
fuse_writer.rs:52) — re-designed for the MI355X deployment: the FUSE daemon
embeds the local GPU's worker, so READ on an HBM-cached block is a
synchronous in-process arena read (hipMemcpyAsync through the pinned ring)
with no event-loop hop; metadata ops bridge to the client's asyncio loop.
"""
from __future__ import annotations

import asyncio
import errno
import logging
import os
import stat as stat_m
import threading
import time
from typing import Optional

from curvine_amd import errors as cverr
from curvine_amd.fuse import abi
from curvine_amd.model import FileStatus

log = logging.getLogger("curvine.fuse.ops")


class Node:
    __slots__ = ("id", "parent", "name", "nlookup", "children")

    def __init__(self, id: int, parent: int, name: str):
        self.id = id
        self.parent = parent
        self.name = name
        self.nlookup = 0
        self.children: dict[str, int] = {}


class FileHandle:
    __slots__ = ("fh", "node_id", "path", "status", "reader", "writer",
                 "write_pos", "flags", "dir_entries", "lock", "nw_registered")

    def __init__(self, fh: int, node_id: int, path: str):
        self.fh = fh
        self.node_id = node_id
        self.path = path
        self.status: Optional[FileStatus] = None
        self.reader = None          # SyncReadState
        self.writer = None          # FsWriter (async, bridged)
        self.write_pos = 0
        self.flags = 0
        self.dir_entries: Optional[list] = None
        self.lock = threading.Lock()
        self.nw_registered = False   # native write window active


class SyncReadState:
    """Per-handle block readers resolved once at open; in-process blocks are
    read synchronously straight from the worker's store."""

    def __init__(self, fs: "CurvineFuseFs", file_blocks):
        self.fs = fs
        self.fb = file_blocks
        self.length = file_blocks.status.length
        self.block_size = file_blocks.status.block_size
        self._offs = None                     # block offset index (lazy)
        self._local: dict[int, object] = {}   # block idx -> store reader
        self._remote = None                   # lazy async FsReader
        # pin every colocated block NOW: store reader refcounts defer
        # deletion, so an open fd keeps serving after unlink/truncate
        # even when the kernel page cache absorbed the early reads
        # (writeback_cache) and the first READ op arrives late
        for idx, lb in enumerate(file_blocks.blocks):
            try:
                self._open_local(idx, lb)
            except Exception:  # noqa: BLE001 — remote/raced blocks: lazy
                pass

    def read_into(self, off: int, out, out_off: int, n: int) -> int:
        import bisect

        from curvine_amd.client.reader import _hole_span
        n = max(0, min(n, self.length - off))
        if self._offs is None:
            self._offs = [b.offset for b in self.fb.blocks]
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
            if idx < 0 or idx >= len(self.fb.blocks):
                break
            lb = self.fb.blocks[idx]
            boff = pos - lb.offset
            want = min(n - got, lb.block.length - boff)
            r = self._local.get(idx)
            if r is None:
                r = self._open_local(idx, lb)
            if r is not None:
                got += r.read_into(boff, out, out_off + got, want)
            else:
                got += self._read_remote(pos, out, out_off + got, want)
        return got

    def _open_local(self, idx: int, lb):
        from curvine_amd.worker import registry
        for addr in lb.locations:
            store = registry.lookup(addr.worker_id)
            if store is not None:
                try:
                    r = store.open_reader(lb.block.block_id)
                    self._local[idx] = r
                    return r
                except Exception:  # noqa: BLE001
                    continue
        # colocated worker in ANOTHER process (the production daemon
        # shape): hipIpc-map its HBM arena for direct DMA reads — the
        # same native read path as the in-process registry hit
        try:
            from curvine_amd.client.block_client import open_ipc_reader
            for addr in lb.locations:
                r = self.fs.call(open_ipc_reader(
                    self.fs.fs.client, addr, lb.block.block_id), 30)
                if r is not None:
                    self._local[idx] = r
                    return r
        except Exception:  # noqa: BLE001 — fall back to the remote path
            pass
        return None

    def read_into_px(self, off: int, ptr: int, view, n: int) -> int:
        """read_into variant for a pinned destination buffer: HBM-resident
        blocks DMA directly to `ptr` (hipMemcpyAsync D2H onto a pooled
        stream); file/remote blocks fall back to the view."""
        import bisect
        import ctypes

        from curvine_amd.client.reader import _hole_span
        n = max(0, min(n, self.length - off))
        if self._offs is None:
            self._offs = [b.offset for b in self.fb.blocks]
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
            if idx < 0 or idx >= len(self.fb.blocks):
                break
            lb = self.fb.blocks[idx]
            boff = pos - lb.offset
            want = min(n - got, lb.block.length - boff)
            r = self._local.get(idx)
            if r is None:
                r = self._open_local(idx, lb)
            if r is not None and r.meta.get("kind") == "arena":
                got += r.read_to_ptr(boff, ptr + got, want, False)
            elif r is not None:
                got += r.read_into(boff, view, got, want)
            else:
                got += self._read_remote(pos, view, got, want)
        return got

    def _read_remote(self, off: int, out, out_off: int, n: int) -> int:
        # remote / hole blocks: bridge to the asyncio FsReader
        if self._remote is None:
            from curvine_amd.client.reader import FsReader
            self._remote = FsReader(self.fs.fs.client, self.fb)
        buf = bytearray(n)
        got = self.fs.call(self._remote.pread_into(off, buf, 0, n))
        out[out_off:out_off + got] = memoryview(buf)[:got]
        return got

    def close(self) -> None:
        for r in self._local.values():
            try:
                r.close()
            except Exception:  # noqa: BLE001
                pass
        self._local.clear()
        if self._remote is not None:
            self._remote.close()


class SyncWriteState:
    """Per-handle writer that keeps the hot path in the channel thread.

    Cross-thread hops (run_coroutine_threadsafe) cost ~1-2 ms in this
    environment, so per-WRITE asyncio bridging caps FUSE writes at tens of
    MB/s.  Instead: the async loop is consulted once per BLOCK (add_block
    RPC); the bytes go synchronously into the colocated worker's store
    (HBM arena / host memcpy, GIL released in C++).  A block placed on a
    non-local worker falls back to a bridged remote stream write.
    """

    def __init__(self, fs: "CurvineFuseFs", status, existing_blocks=None):
        self.fs = fs
        self.status = status
        self.path = status.path
        self.block_size = status.block_size
        self.pos = status.length if existing_blocks else 0
        self._block_lens = [b.block.length for b in (existing_blocks or [])]
        # (block_id, locations, tiers) per settled block — random writes
        # reopen these for in-place rewrite
        self._block_addrs = [(b.block.block_id, b.locations, b.tiers)
                             for b in (existing_blocks or [])]
        self._rw: dict[int, list] = {}   # block idx -> rewrite writers
        self._commits: list[dict] = []
        self._cur = None           # sync store BlockWriter
        self._cur_store = None
        self._cur_lb = None
        self._cur_async = None     # fallback async writers
        self._cur_pos = 0
        self._done = False
        # shared-writer state (backend_handle.rs:288-310 analog): several
        # FUSE handles may hold this writer concurrently
        import threading as _th
        self.wlock = _th.Lock()
        self.refs = 1
        self.native_handle = None   # FileHandle owning a native window

    def write(self, data, ptr: int | None = None) -> int:
        """`ptr`, when given, is the raw address of `data` inside a pinned
        buffer — HBM-bound bytes then DMA host->device directly."""
        data = memoryview(data)
        total = len(data)
        consumed = 0
        while consumed < total:
            if self._cur is None and self._cur_async is None:
                self._next_block()
            room = self.block_size - self._cur_pos
            take = min(room, total - consumed)
            chunk = data[consumed:consumed + take]
            if self._cur is not None:
                if ptr is not None and self._cur.meta.get("kind") == "arena":
                    self._cur.write_from_ptr(ptr + consumed, take, False)
                else:
                    self._cur.write(chunk, take)
            else:
                payload = bytes(chunk)
                self.fs.call(_gather_writes(self._cur_async, payload))
            self._cur_pos += take
            consumed += take
            if self._cur_pos >= self.block_size:
                self._commit_block()
        self.pos += total
        return total

    def _next_block(self) -> None:
        from curvine_amd.worker import registry
        lb = self.fs.call(self.fs.fs.client.add_block(self.path))
        self._cur_lb = lb
        self._cur_pos = 0
        if len(lb.locations) == 1:
            store = registry.lookup(lb.locations[0].worker_id)
            if store is not None:
                self._cur_store = store
                self._cur = store.create_writer(
                    lb.block.block_id, self.block_size, lb.tiers[0])
                return
        # replicated or remote placement: bridged async star write
        from curvine_amd.client.block_client import make_block_writer
        self._cur_async = [make_block_writer(a, lb.block.block_id,
                                             self.block_size, t)
                           for a, t in zip(lb.locations, lb.tiers)]

    def _commit_block(self) -> None:
        lb = self._cur_lb
        if self._cur is not None:
            tier = self._cur_store.finalize(lb.block.block_id, self._cur_pos)
            tiers = [tier]
        else:
            tiers = self.fs.call(_gather_commits(self._cur_async, self._cur_pos))
        self._commits.append({
            "block_id": lb.block.block_id,
            "locations": [a.worker_id for a in lb.locations],
            "tiers": [t or hint for t, hint in zip(tiers, lb.tiers)]})
        self._block_lens.append(self._cur_pos)
        self._block_addrs.append((lb.block.block_id, list(lb.locations),
                                  [t or hint for t, hint
                                   in zip(tiers, lb.tiers)]))
        self._cur = self._cur_store = self._cur_lb = self._cur_async = None
        self._cur_pos = 0

    # ---------------- random (backward) writes ----------------
    def pwrite_back(self, off: int, data) -> int:
        """In-place rewrite of already-written bytes [off, off+len) with
        off+len <= self.pos (fs_writer_base.rs seek-write analog).
        Settled blocks are reopened for positional rewrite on every
        replica; the current open block is patched directly."""
        data = memoryview(data)
        n = len(data)
        if off + n > self.pos:
            raise cverr.OutOfRange(
                f"rewrite [{off},{off + n}) past {self.pos}")
        cur_start = sum(self._block_lens)
        consumed = 0
        while consumed < n:
            o = off + consumed
            if o >= cur_start:                     # current open block
                boff = o - cur_start
                take = min(n - consumed, self._cur_pos - boff)
                chunk = data[consumed:consumed + take]
                if self._cur is not None:
                    self._cur.pwrite(boff, chunk, take)
                else:
                    self.fs.call(_gather_pwrites(
                        self._cur_async, boff, bytes(chunk)))
            else:                                  # settled block
                import bisect
                starts = []
                s = 0
                for ln in self._block_lens:
                    starts.append(s)
                    s += ln
                idx = bisect.bisect_right(starts, o) - 1
                boff = o - starts[idx]
                take = min(n - consumed, self._block_lens[idx] - boff)
                chunk = data[consumed:consumed + take]
                ws = self._rewrite_writers(idx)
                if len(ws) == 1 and hasattr(ws[0], "writer"):
                    ws[0].writer.pwrite(boff, chunk, take)   # local, no hop
                else:
                    self.fs.call(_gather_pwrites(ws, boff, bytes(chunk)))
            consumed += take
        return n

    def _rewrite_writers(self, idx: int) -> list:
        ws = self._rw.get(idx)
        if ws is None:
            from curvine_amd.client.block_client import make_block_writer
            bid, addrs, tiers = self._block_addrs[idx]
            ws = [make_block_writer(a, bid, 0, t, reopen=True)
                  for a, t in zip(addrs, tiers)]
            self._rw[idx] = ws
        return ws

    _ZEROS = bytes(1 << 20)

    def advance(self, n: int) -> None:
        """Account bytes appended directly into the current block by the
        native FUSE loop (the window never exceeds the block boundary)."""
        self._cur_pos += n
        self.pos += n
        if self._cur_pos >= self.block_size:
            self._commit_block()

    def write_zeros(self, n: int) -> int:
        """Sparse forward seek: fill the hole with zeros
        (sparse_hole_flush_read_test analog)."""
        remaining = n
        while remaining > 0:
            take = min(len(self._ZEROS), remaining)
            self.write(memoryview(self._ZEROS)[:take])
            remaining -= take
        return n

    def flush(self) -> None:
        pass   # partial blocks are committed at complete()

    def complete(self):
        if self._done:
            return self.status
        if self._cur is not None or self._cur_async is not None:
            self._commit_block()
        if self._rw:
            # close rewrite streams (no finalize: blocks stay as-is)
            for ws in self._rw.values():
                self.fs.call(_gather_commits(ws, None))
            self._rw = {}
        self._done = True
        length = sum(self._block_lens)
        st = self.fs.call(self.fs.fs.client.complete_file(
            self.path, length, self._block_lens, self._commits))
        self.status = st
        return st

    def abort(self) -> None:
        self._done = True
        if self._cur is not None and self._cur_store is not None:
            self._cur_store.abort(self._cur_lb.block.block_id)


async def _gather_writes(writers, payload):
    import asyncio
    await asyncio.gather(*[w.write(payload) for w in writers])


async def _gather_pwrites(writers, off, payload):
    import asyncio
    await asyncio.gather(*[w.pwrite(off, payload) for w in writers])


async def _gather_commits(writers, length):
    import asyncio
    return list(await asyncio.gather(*[w.commit(length) for w in writers]))


class CurvineFuseFs:
    """Opcode handler table over CurvineFileSystem."""

    def __init__(self, fs, conf, loop: asyncio.AbstractEventLoop):
        self.fs = fs               # CurvineFileSystem (async)
        self.conf = conf
        self.loop = loop
        self.session = None
        self.nodes: dict[int, Node] = {1: Node(1, 0, "")}
        self.nodes_lock = threading.Lock()
        self.next_node = 2
        self.handles: dict[int, FileHandle] = {}
        # node_id -> live SyncWriteState shared across handles
        self.shared_writers: dict[int, "SyncWriteState"] = {}
        # serializes write-open registry check+create across channels
        self.wopen_lock = threading.Lock()
        self.next_fh = 1
        self.handles_lock = threading.Lock()
        self.attr_cache: dict[int, tuple[FileStatus, float]] = {}
        self.attr_ttl = conf.fuse.attr_ttl_ms / 1000.0
        self.entry_ttl = conf.fuse.entry_ttl_ms / 1000.0
        # POSIX advisory locks: (node_id) -> list of (start, end, type, owner, pid)
        self.plocks: dict[int, list] = {}
        self.plock_mu = threading.Lock()
        # kernel INTERRUPT targets: uniques of in-flight requests the
        # kernel asked to abort (consumed by blocking SETLKW polls)
        self.interrupted: set[int] = set()

    # ---------------- helpers ----------------
    def call(self, coro, timeout: float = 120.0):
        return asyncio.run_coroutine_threadsafe(coro, self.loop).result(timeout)

    def node_path(self, node_id: int) -> str:
        with self.nodes_lock:
            parts = []
            node = self.nodes.get(node_id)
            if node is None:
                raise OSError(errno.ESTALE, "stale nodeid")
            while node.id != 1:
                parts.append(node.name)
                node = self.nodes[node.parent]
            return "/" + "/".join(reversed(parts))

    def child_node(self, parent_id: int, name: str) -> Node:
        with self.nodes_lock:
            parent = self.nodes[parent_id]
            nid = parent.children.get(name)
            if nid is not None:
                return self.nodes[nid]
            node = Node(self.next_node, parent_id, name)
            self.next_node += 1
            self.nodes[node.id] = node
            parent.children[name] = node.id
            return node

    def drop_child(self, parent_id: int, name: str) -> None:
        with self.nodes_lock:
            parent = self.nodes.get(parent_id)
            if parent:
                parent.children.pop(name, None)

    def cached_status(self, node_id: int) -> Optional[FileStatus]:
        ent = self.attr_cache.get(node_id)
        if ent and ent[1] > time.monotonic():
            return ent[0]
        return None

    def cache_status(self, node_id: int, st: FileStatus) -> None:
        self.attr_cache[node_id] = (st, time.monotonic() + self.attr_ttl)

    def invalidate(self, node_id: int) -> None:
        self.attr_cache.pop(node_id, None)

    def stat_path(self, node_id: int, path: str) -> FileStatus:
        st = self.cached_status(node_id)
        if st is None:
            st = self.call(self.fs.file_status(path))
            self.cache_status(node_id, st)
        return st

    def entry_out(self, node: Node, st: FileStatus) -> bytes:
        node.nlookup += 1
        self.cache_status(node.id, st)
        attr = abi.pack_attr(self._with_ino(st, node.id))
        ev = int(self.entry_ttl)
        evn = int((self.entry_ttl - ev) * 1e9)
        return abi.ENTRY_OUT.pack(node.id, 0, ev, ev, evn, evn) + attr

    def attr_out(self, node_id: int, st: FileStatus) -> bytes:
        av = int(self.attr_ttl)
        avn = int((self.attr_ttl - av) * 1e9)
        return abi.ATTR_OUT.pack(av, avn, 0) + abi.pack_attr(
            self._with_ino(st, node_id))

    @staticmethod
    def _with_ino(st: FileStatus, node_id: int) -> FileStatus:
        # displayed st_ino = the MASTER's inode id (hardlinked names show
        # the same ino); the FUSE nodeid (protocol handle) stays ours
        if st.inode_id:
            return st
        import copy
        st2 = copy.copy(st)
        st2.inode_id = node_id
        return st2

    # ---------------- native write-window plumbing ----------------
    def _sync_native_write(self, h: FileHandle) -> None:
        """Fold natively-appended bytes back into the Python writer state
        (destructive: the window is removed; re-registered after the next
        Python-side write)."""
        if not h.nw_registered or self.session is None or \
                self.session.native_id is None:
            return
        from curvine_amd.native import load
        n = load().fuse_loop_unregister_write(self.session.native_id, h.fh)
        h.nw_registered = False
        if h.writer is not None and h.writer.native_handle is h:
            h.writer.native_handle = None
        if n and h.writer is not None:
            h.writer.advance(n)
            h.write_pos += n

    def _native_write_extra(self, h: FileHandle) -> int:
        if not h.nw_registered or self.session is None or \
                self.session.native_id is None:
            return 0
        from curvine_amd.native import load
        return load().fuse_loop_write_state(self.session.native_id, h.fh)

    def _register_native_write(self, h: FileHandle) -> None:
        w = h.writer
        if self.fs.conf.fuse.writeback_cache:
            # kernel writeback flushes pages concurrently and not always
            # in order; the native CAS-append window assumes a single
            # in-order WRITE stream, so keep writes on the serialized
            # Python path in this mode
            return
        if self.session is None or self.session.native_id is None or \
                w is None or w._done or w._cur is None or \
                w.refs > 1 or w._cur.meta.get("kind") != "arena":
            return
        room = w.block_size - w._cur_pos
        if room <= 0:
            return
        from curvine_amd.native import load
        load().fuse_loop_register_write(
            self.session.native_id, h.fh, w._cur.layout.arena.handle,
            w._cur.meta["offset"] + w._cur_pos, room, h.write_pos)
        h.nw_registered = True
        w.native_handle = h

    def new_handle(self, node_id: int, path: str) -> FileHandle:
        with self.handles_lock:
            fh = self.next_fh
            self.next_fh += 1
            h = FileHandle(fh, node_id, path)
            self.handles[fh] = h
            return h

    def get_handle(self, fh: int) -> FileHandle:
        h = self.handles.get(fh)
        if h is None:
            raise OSError(errno.EBADF, f"fh {fh}")
        return h

    # ---------------- ops ----------------
    def op_init(self, nodeid, body, ctx):
        major, minor, max_readahead, flags = abi.INIT_IN.unpack_from(body, 0)
        log.info("fuse INIT kernel %d.%d flags=%#x", major, minor, flags)
        out_flags = (abi.FUSE_ASYNC_READ | abi.FUSE_BIG_WRITES |
                     abi.FUSE_PARALLEL_DIROPS | abi.FUSE_ATOMIC_O_TRUNC |
                     abi.FUSE_MAX_PAGES | abi.FUSE_CACHE_SYMLINKS |
                     abi.FUSE_HANDLE_KILLPRIV_V2 |
                     abi.FUSE_DO_READDIRPLUS | abi.FUSE_READDIRPLUS_AUTO |
                     abi.FUSE_POSIX_LOCKS) & flags | abi.FUSE_MAX_PAGES
        if self.conf.fuse.writeback_cache:
            out_flags |= abi.FUSE_WRITEBACK_CACHE & flags
        max_write = self.conf.fuse.max_write
        return abi.INIT_OUT.pack(
            7, min(minor, abi.FUSE_KERNEL_MINOR_VERSION),
            self.conf.fuse.max_readahead, out_flags,
            64, 48, max_write, 1,
            max(1, max_write // 4096), 0, 0, 0)

    def op_destroy(self, nodeid, body, ctx):
        return b""

    def op_lookup(self, nodeid, body, ctx):
        name = bytes(body).split(b"\x00", 1)[0].decode()
        parent_path = self.node_path(nodeid)
        path = (parent_path.rstrip("/") + "/" + name)
        try:
            node = self.child_node(nodeid, name)
            st = self.stat_path(node.id, path)
        except cverr.FileNotFound:
            raise OSError(errno.ENOENT, path)
        return self.entry_out(node, st)

    def op_forget(self, nodeid, body, ctx):
        (nlookup,) = abi.FORGET_IN.unpack_from(body, 0)
        self._forget_one(nodeid, nlookup)
        return None

    def op_batch_forget(self, nodeid, body, ctx):
        count, _ = abi.BATCH_FORGET_IN.unpack_from(body, 0)
        off = abi.BATCH_FORGET_IN.size
        for _ in range(count):
            nid, nl = abi.FORGET_ONE.unpack_from(body, off)
            off += abi.FORGET_ONE.size
            self._forget_one(nid, nl)
        return None

    def _forget_one(self, nid: int, nlookup: int) -> None:
        with self.nodes_lock:
            node = self.nodes.get(nid)
            if node is None or nid == 1:
                return
            node.nlookup -= nlookup
            if node.nlookup <= 0 and not node.children:
                self.nodes.pop(nid, None)
                parent = self.nodes.get(node.parent)
                if parent and parent.children.get(node.name) == nid:
                    parent.children.pop(node.name, None)
                self.attr_cache.pop(nid, None)

    def op_getattr(self, nodeid, body, ctx):
        path = self.node_path(nodeid)
        try:
            st = self.stat_path(nodeid, path)
        except cverr.FileNotFound:
            # fstat of an open-but-unlinked (or renamed-away) file: POSIX
            # serves the inode through the handle (the kernel revalidates
            # attrs before cached reads under writeback_cache)
            st = None
            with self.handles_lock:
                for h in self.handles.values():
                    if h.node_id == nodeid and h.status is not None:
                        st = h.status
                        break
            if st is None:
                raise OSError(errno.ENOENT, path)
        # live write handle: report current write position as size
        st = self._adjust_writing_size(nodeid, st)
        return self.attr_out(nodeid, st)

    def _adjust_writing_size(self, nodeid: int, st: FileStatus) -> FileStatus:
        with self.handles_lock:
            for h in self.handles.values():
                if h.node_id == nodeid and h.writer is not None:
                    import copy
                    st2 = copy.copy(st)
                    # the SHARED append end can be past this handle's own
                    # position (another handle appended, or an extending
                    # truncate zero-filled through the shared writer)
                    st2.length = max(st.length,
                                     h.write_pos + self._native_write_extra(h),
                                     getattr(h.writer, "pos", 0))
                    return st2
        return st

    def op_setattr(self, nodeid, body, ctx):
        (valid, _pad, fh, size, _lo, at, mt, ct, atn, mtn, ctn, mode,
         _u4, uid, gid, _u5) = abi.SETATTR_IN.unpack_from(body, 0)
        path = self.node_path(nodeid)
        attrs = {}
        if valid & abi.FATTR_MODE:
            attrs["mode"] = mode & 0o7777
        if valid & abi.FATTR_UID:
            attrs["uid"] = uid
        if valid & abi.FATTR_GID:
            attrs["gid"] = gid
        if valid & (abi.FATTR_ATIME | abi.FATTR_ATIME_NOW):
            attrs["atime_ms"] = int(time.time() * 1000) \
                if valid & abi.FATTR_ATIME_NOW else at * 1000 + atn // 1_000_000
        if valid & (abi.FATTR_MTIME | abi.FATTR_MTIME_NOW):
            attrs["mtime_ms"] = int(time.time() * 1000) \
                if valid & abi.FATTR_MTIME_NOW else mt * 1000 + mtn // 1_000_000
        self.invalidate(nodeid)
        try:
            if valid & abi.FATTR_SIZE:
                st = self._truncate(nodeid, path, size,
                                    fh if valid & abi.FATTR_FH else 0)
            if attrs:
                st = self.call(self.fs.set_attr(path, **attrs))
            else:
                st = self.stat_path(nodeid, path)
        except (cverr.FileNotFound, OSError) as e:
            if isinstance(e, OSError) and e.errno != errno.ENOENT:
                raise
            # SETATTR on an open-but-unlinked node: the kernel flushes
            # timestamps through write_inode (writeback_cache) at close;
            # failing it would poison the mapping and turn every later
            # close() into EIO.  Serve from the handle like getattr does.
            st = None
            with self.handles_lock:
                for h in self.handles.values():
                    if h.node_id == nodeid and h.status is not None:
                        st = h.status
                        break
            if st is None:
                raise
            if "mtime_ms" in attrs:
                st.mtime_ms = attrs["mtime_ms"]
            if "atime_ms" in attrs:
                st.atime_ms = attrs["atime_ms"]
        st = self._adjust_writing_size(nodeid, st)
        self.cache_status(nodeid, st)
        return self.attr_out(nodeid, st)

    def _truncate(self, nodeid, path, size, fh) -> FileStatus:
        # open write handle truncating to its own position: no-op
        if fh:
            h = self.handles.get(fh)
            if h is not None and h.writer is not None and h.write_pos == size:
                return self.stat_path(nodeid, path)
        try:
            cur = self.call(self.fs.file_status(path))
        except cverr.FileNotFound:
            raise OSError(errno.ENOENT, path)
        if size == cur.length:
            return cur
        if size < cur.length:
            return self.call(self.fs.resize(path, size))
        # extending truncate (ftruncate growth / posix_fallocate): with
        # an open writer extend its sparse watermark so close() keeps
        # the new size; otherwise grow the metadata length — the tail
        # reads back as a hole (BlockReaderHole zeros)
        with self.handles_lock:
            shared = self.shared_writers.get(nodeid)
        if shared is not None:
            with shared.wlock:
                nh = shared.native_handle
                if nh is not None:
                    self._sync_native_write(nh)
                if size > shared.pos:
                    shared.write_zeros(size - shared.pos)
            return self.stat_path(nodeid, path)
        return self.call(self.fs.resize(path, size))

    def op_mkdir(self, nodeid, body, ctx):
        mode, _umask = abi.MKDIR_IN.unpack_from(body, 0)
        name = bytes(body[abi.MKDIR_IN.size:]).split(b"\x00", 1)[0].decode()
        path = self.node_path(nodeid).rstrip("/") + "/" + name
        st = self.call(self.fs.mkdir(path, mode & 0o7777, create_parents=False))
        node = self.child_node(nodeid, name)
        return self.entry_out(node, st)

    def op_unlink(self, nodeid, body, ctx):
        name = bytes(body).split(b"\x00", 1)[0].decode()
        path = self.node_path(nodeid).rstrip("/") + "/" + name
        self.call(self.fs.delete(path, recursive=False))
        self.drop_child(nodeid, name)
        return b""

    def op_rmdir(self, nodeid, body, ctx):
        name = bytes(body).split(b"\x00", 1)[0].decode()
        path = self.node_path(nodeid).rstrip("/") + "/" + name
        st = self.call(self.fs.file_status(path))
        if not st.is_dir:
            raise OSError(errno.ENOTDIR, path)
        self.call(self.fs.delete(path, recursive=False))
        self.drop_child(nodeid, name)
        return b""

    def _rewrite_handle_paths(self, old: str, new: str) -> None:
        """Retarget open handles after a rename (covers files and whole
        directory subtrees)."""
        with self.handles_lock:
            for h in self.handles.values():
                if h.path == old or h.path.startswith(old + "/"):
                    h.path = new + h.path[len(old):]
                    if h.writer is not None:
                        h.writer.path = h.path

    def _rename(self, nodeid, newdir, oldname, newname, flags=0):
        src = self.node_path(nodeid).rstrip("/") + "/" + oldname
        dst = self.node_path(newdir).rstrip("/") + "/" + newname
        if flags & 1:   # RENAME_NOREPLACE
            if self.call(self.fs.exists(dst)):
                raise OSError(errno.EEXIST, dst)
        if flags & 2:   # RENAME_EXCHANGE
            tmp = dst + f".xchg.{time.monotonic_ns()}"
            self.call(self.fs.rename(dst, tmp))
            self.call(self.fs.rename(src, dst))
            self.call(self.fs.rename(tmp, src))
            # the kernel swaps the dentry<->inode associations, so swap our
            # nodeid<->path mapping the same way: each nodeid keeps naming
            # its (moved) content and page caches stay coherent
            with self.nodes_lock:
                sp = self.nodes.get(nodeid)
                dp = self.nodes.get(newdir)
                na = sp.children.get(oldname) if sp else None
                nb = dp.children.get(newname) if dp else None
                if na is not None and nb is not None:
                    sp.children[oldname] = nb
                    dp.children[newname] = na
                    self.nodes[na].parent = newdir
                    self.nodes[na].name = newname
                    self.nodes[nb].parent = nodeid
                    self.nodes[nb].name = oldname
            for nid in (na, nb):
                if nid is not None:
                    self.invalidate(nid)
        else:
            self.call(self.fs.rename(src, dst))
            # open handles are path-keyed (add_block/complete_file RPCs):
            # retarget them so a write fd keeps working under the new name
            self._rewrite_handle_paths(src, dst)
        with self.nodes_lock:
            src_parent = self.nodes.get(nodeid)
            nid = src_parent.children.pop(oldname, None) if src_parent else None
            if nid is not None and not flags & 2:
                node = self.nodes[nid]
                node.parent = newdir
                node.name = newname
                dst_parent = self.nodes.get(newdir)
                if dst_parent is not None:
                    old = dst_parent.children.get(newname)
                    if old is not None:
                        self.attr_cache.pop(old, None)
                    dst_parent.children[newname] = nid
                self.attr_cache.pop(nid, None)
        return b""

    def op_rename(self, nodeid, body, ctx):
        (newdir,) = abi.RENAME_IN.unpack_from(body, 0)
        names = bytes(body[abi.RENAME_IN.size:]).split(b"\x00")
        return self._rename(nodeid, newdir, names[0].decode(), names[1].decode())

    def op_rename2(self, nodeid, body, ctx):
        newdir, flags, _ = abi.RENAME2_IN.unpack_from(body, 0)
        names = bytes(body[abi.RENAME2_IN.size:]).split(b"\x00")
        return self._rename(nodeid, newdir, names[0].decode(),
                            names[1].decode(), flags)

    def op_symlink(self, nodeid, body, ctx):
        parts = bytes(body).split(b"\x00")
        name, target = parts[0].decode(), parts[1].decode()
        path = self.node_path(nodeid).rstrip("/") + "/" + name
        st = self.call(self.fs.symlink(path, target))
        node = self.child_node(nodeid, name)
        return self.entry_out(node, st)

    def op_readlink(self, nodeid, body, ctx):
        st = self.stat_path(nodeid, self.node_path(nodeid))
        if not st.is_symlink:
            raise OSError(errno.EINVAL, "not a symlink")
        return st.symlink_target.encode()

    def op_link(self, nodeid, body, ctx):
        (oldnode,) = abi.LINK_IN.unpack_from(body, 0)
        name = bytes(body[abi.LINK_IN.size:]).split(b"\x00", 1)[0].decode()
        src = self.node_path(oldnode)
        dst = self.node_path(nodeid).rstrip("/") + "/" + name
        st = self.call(self.fs.link(src, dst))
        node = self.child_node(nodeid, name)
        self.invalidate(oldnode)
        return self.entry_out(node, st)

    def op_mknod(self, nodeid, body, ctx):
        mode, rdev, _umask, _ = abi.MKNOD_IN.unpack_from(body, 0)
        if not stat_m.S_ISREG(mode):
            raise OSError(errno.ENOTSUP, "only regular files")
        name = bytes(body[abi.MKNOD_IN.size:]).split(b"\x00", 1)[0].decode()
        path = self.node_path(nodeid).rstrip("/") + "/" + name
        self.call(self.fs.client.create(path, overwrite=False,
                                        mode=mode & 0o7777))
        st = self.call(self.fs.client.complete_file(path, 0, []))
        node = self.child_node(nodeid, name)
        return self.entry_out(node, st)

    # ---------------- open/create/read/write ----------------
    def op_create(self, nodeid, body, ctx):
        flags, mode, _umask, _of = abi.CREATE_IN.unpack_from(body, 0)
        name = bytes(body[abi.CREATE_IN.size:]).split(b"\x00", 1)[0].decode()
        path = self.node_path(nodeid).rstrip("/") + "/" + name
        excl = bool(flags & os.O_EXCL)
        st = self.call(self.fs.client.create(path, overwrite=not excl,
                                             mode=mode & 0o7777))
        h = self.new_handle(0, path)
        h.writer = SyncWriteState(self, st)
        h.flags = flags
        h.status = st
        node = self.child_node(nodeid, name)
        h.node_id = node.id
        # register like op_open's write branch: later write-opens (and
        # extending truncate) must find this live writer
        with self.handles_lock:
            self.shared_writers[node.id] = h.writer
        self.invalidate(node.id)
        entry = self.entry_out(node, st)
        open_out = abi.OPEN_OUT.pack(h.fh, 0, 0)
        return entry + open_out

    def op_open(self, nodeid, body, ctx):
        flags, _ = abi.OPEN_IN.unpack_from(body, 0)
        path = self.node_path(nodeid)
        accmode = flags & os.O_ACCMODE
        h = self.new_handle(nodeid, path)
        h.flags = flags
        try:
            if accmode == os.O_RDONLY:
                fb = self.call(self.fs.client.open(path))
                h.status = fb.status
                h.reader = SyncReadState(self, fb)
                if self.session is not None:
                    # HBM/host-arena-resident files: READs served by the
                    # GIL-free native loop from here on
                    try:
                        self.session.try_register_read(h.fh, h.reader)
                    except Exception as e:  # noqa: BLE001
                        log.debug("native read registration failed: %s", e)
            else:
                # the registry check + append-lease acquisition must be
                # atomic across FUSE channels, or two simultaneous
                # write-opens both miss and the loser gets FileInWriting
                self.wopen_lock.acquire()
                try:
                    shared = None
                    if not flags & os.O_TRUNC:
                        # POSIX allows several concurrent write-opens of
                        # one file: share the live writer instead of
                        # failing the append lease (shared-writer
                        # semantics, backend_handle.rs:288-310)
                        with self.handles_lock:
                            sw = self.shared_writers.get(nodeid)
                            if sw is not None and not sw._done:
                                sw.refs += 1
                                shared = sw
                    self._wopen_branch(h, nodeid, path, flags, accmode,
                                       shared)
                finally:
                    self.wopen_lock.release()
                h.status = h.writer.status
                if accmode == os.O_RDWR:
                    try:
                        fb = self.call(self.fs.client.open(path))
                        h.reader = SyncReadState(self, fb)
                    except cverr.FsError:
                        pass
            self.invalidate(nodeid)
            return abi.OPEN_OUT.pack(h.fh, 0, 0)
        except Exception:
            with self.handles_lock:
                self.handles.pop(h.fh, None)
            raise

    def _wopen_branch(self, h, nodeid, path, flags, accmode, shared):
        import os
        if shared is not None:
            nh = shared.native_handle
            if nh is not None:
                with nh.lock:
                    self._sync_native_write(nh)
            h.writer = shared
            h.write_pos = shared.pos
        elif flags & os.O_TRUNC:
            st = self.call(self.fs.client.create(path, overwrite=True))
            h.writer = SyncWriteState(self, st)
            h.write_pos = 0
            with self.handles_lock:
                self.shared_writers[nodeid] = h.writer
        elif flags & os.O_APPEND or accmode in (os.O_WRONLY, os.O_RDWR):
            st = self.call(self.fs.file_status(path))
            if st.length == 0:
                st = self.call(self.fs.client.create(path, overwrite=True))
                h.writer = SyncWriteState(self, st)
            else:
                fb = self.call(self.fs.client.append(path))
                h.writer = SyncWriteState(self, fb.status, fb.blocks)
            h.write_pos = h.writer.pos
            with self.handles_lock:
                self.shared_writers[nodeid] = h.writer

    def op_read(self, nodeid, body, ctx):
        fh, offset, size, _rf, _lo, _fl, _ = abi.READ_IN.unpack_from(body, 0)
        h = self.get_handle(fh)
        if h.reader is None and h.writer is None:
            # restored handle (hot upgrade): rebuild the reader lazily
            fb = self.call(self.fs.client.open(h.path))
            h.status = fb.status
            h.reader = SyncReadState(self, fb)
        if h.reader is None:
            # O_WRONLY handle read: under writeback_cache the kernel
            # reads through ANY open handle to prefill partial pages
            # before writing them back — serve the settled content
            with h.lock:
                self._sync_native_write(h)
            try:
                fb = self.call(self.fs.client.open(h.path))
                h.reader = SyncReadState(self, fb)
            except Exception:  # noqa: BLE001
                raise OSError(errno.EBADF, "not open for read")
        ch = ctx[4]
        pin = getattr(ch, "reply_pin", None)
        if pin is not None and size <= pin.nbytes - 64 and \
                hasattr(h.reader, "read_into_px"):
            n = h.reader.read_into_px(offset, pin.ptr, pin.view, size)
            return pin.view[:n]
        buf = bytearray(size)
        n = h.reader.read_into(offset, buf, 0, size)
        return memoryview(buf)[:n]

    def op_write(self, nodeid, body, ctx):
        fh, offset, size, _wf, _lo, _fl, _ = abi.WRITE_IN.unpack_from(body, 0)
        h = self.get_handle(fh)
        data = body[abi.WRITE_IN.size:abi.WRITE_IN.size + size]
        ch = ctx[4]
        req = getattr(ch, "req_buf", None)
        ptr = (req.ptr + abi.IN_HEADER_SIZE + abi.WRITE_IN.size
               if req is not None else None)
        with h.lock:
            self._sync_native_write(h)
            if h.writer is None:
                raise OSError(errno.EBADF, "not open for write")
            if h.writer._done:
                # a dup'd fd wrote after an earlier close flushed the file:
                # transparently reopen for append
                fb = self.call(self.fs.client.append(h.path))
                h.writer = SyncWriteState(self, fb.status, fb.blocks)
                h.write_pos = h.writer.pos
                with self.handles_lock:
                    self.shared_writers[h.node_id] = h.writer
            w = h.writer
            with w.wlock:
                nh = w.native_handle
                if nh is not None and nh is not h:
                    # another handle has an open native append window:
                    # fold it before touching the shared position
                    self._sync_native_write(nh)
                end = w.pos   # the SHARED append end (several handles
                #               may interleave; h.write_pos is per-handle)
                if offset > end:
                    # forward seek: zero-fill the sparse hole
                    w.write_zeros(offset - end)
                elif offset < end:
                    # random write: rewrite the overlap in place, append
                    # any tail past the current end
                    overlap = min(size, end - offset)
                    w.pwrite_back(offset, data[:overlap])
                    if overlap < size:
                        w.write(data[overlap:])
                    h.write_pos = offset + size
                    self._register_native_write(h)
                    return abi.WRITE_OUT.pack(size, 0)
                w.write(data, ptr=ptr)
                h.write_pos = w.pos
                # subsequent sequential WRITEs append GIL-free in C++
                self._register_native_write(h)
        return abi.WRITE_OUT.pack(size, 0)

    def op_flush(self, nodeid, body, ctx):
        """FLUSH completes the file: it is the only write-side op that is
        synchronous with close(2) (RELEASE is async, so completing there
        races with an immediately following OPEN on another channel)."""
        fh, _u, _p, _lo = abi.FLUSH_IN.unpack_from(body, 0)
        h = self.handles.get(fh)
        if h is not None and h.writer is not None:
            with h.lock, h.writer.wlock:
                self._sync_native_write(h)
                if not h.writer._done and h.writer.refs <= 1:
                    # last write handle: complete on close(2).  With
                    # other handles still writing, completion waits for
                    # the final release.
                    st = h.writer.complete()
                    self.cache_status(h.node_id, st)
        return b""

    def op_fsync(self, nodeid, body, ctx):
        return b""

    def op_release(self, nodeid, body, ctx):
        fh, _f, _rf, _lo = abi.RELEASE_IN.unpack_from(body, 0)
        with self.handles_lock:
            h = self.handles.pop(fh, None)
        if h is None:
            return b""
        if h.writer is not None:
            with h.lock, h.writer.wlock:
                self._sync_native_write(h)
                h.writer.refs -= 1
                if h.writer.refs <= 0:
                    if not h.writer._done:
                        st = h.writer.complete()
                        self.cache_status(h.node_id, st)
                    with self.handles_lock:
                        if self.shared_writers.get(h.node_id) is h.writer:
                            del self.shared_writers[h.node_id]
        if h.reader is not None:
            # unregister from the native loop BEFORE closing store readers
            # (registration pins the blocks via reader refcounts)
            if self.session is not None:
                self.session.unregister_read(fh)
            h.reader.close()
        with self.plock_mu:
            self.plocks.pop(fh, None)
        self.invalidate(h.node_id)
        return b""

    # ---------------- dirs ----------------
    def op_opendir(self, nodeid, body, ctx):
        path = self.node_path(nodeid)
        h = self.new_handle(nodeid, path)
        return abi.OPEN_OUT.pack(h.fh, 0, 0)

    def op_readdir(self, nodeid, body, ctx):
        fh, offset, size, _rf, _lo, _fl, _ = abi.READ_IN.unpack_from(body, 0)
        h = self.get_handle(fh)
        if h.dir_entries is None:
            sts = self.call(self.fs.list_status(h.path))
            entries = [(".", None), ("..", None)]
            entries += [(s.name, s) for s in sts]
            h.dir_entries = entries
        out = bytearray()
        idx = offset
        while idx < len(h.dir_entries):
            name, st = h.dir_entries[idx]
            if st is None:
                ino, dtype = nodeid, abi.DT_DIR
            else:
                node = self.child_node(h.node_id, name)
                ino = node.id
                dtype = abi.DT_DIR if st.is_dir else (
                    abi.DT_LNK if st.is_symlink else abi.DT_REG)
            ent = abi.pack_dirent(ino, idx + 1, name.encode(), dtype)
            if len(out) + len(ent) > size:
                break
            out += ent
            idx += 1
        return bytes(out)

    def op_readdirplus(self, nodeid, body, ctx):
        """One round trip for `ls -l`: each entry carries a full
        fuse_entry_out (lookup + attrs) ahead of the dirent
        (fuse_lowlevel readdirplus analog; entries count as lookups)."""
        fh, offset, size, _rf, _lo, _fl, _ = abi.READ_IN.unpack_from(body, 0)
        h = self.get_handle(fh)
        if h.dir_entries is None:
            sts = self.call(self.fs.list_status(h.path))
            entries = [(".", None), ("..", None)]
            entries += [(s.name, s) for s in sts]
            h.dir_entries = entries
        empty_entry = abi.ENTRY_OUT.pack(0, 0, 0, 0, 0, 0) + \
            b"\x00" * abi.ATTR.size
        out = bytearray()
        idx = offset
        while idx < len(h.dir_entries):
            name, st = h.dir_entries[idx]
            if st is None:
                # "." / "..": dirent only, zeroed entry (no lookup taken)
                entry = empty_entry
                ino, dtype = nodeid, abi.DT_DIR
            else:
                node = self.child_node(h.node_id, name)
                entry = self.entry_out(node, st)
                ino = node.id
                dtype = abi.DT_DIR if st.is_dir else (
                    abi.DT_LNK if st.is_symlink else abi.DT_REG)
            ent = entry + abi.pack_dirent(ino, idx + 1, name.encode(), dtype)
            if len(out) + len(ent) > size:
                break
            out += ent
            idx += 1
        return bytes(out)

    def op_releasedir(self, nodeid, body, ctx):
        fh, _f, _rf, _lo = abi.RELEASE_IN.unpack_from(body, 0)
        with self.handles_lock:
            self.handles.pop(fh, None)
        return b""

    def op_fsyncdir(self, nodeid, body, ctx):
        return b""

    # ---------------- misc ----------------
    def op_statfs(self, nodeid, body, ctx):
        info = self.call(self.fs.get_master_info())
        bsize = 4096
        blocks = max(1, info.get("capacity", 0) // bsize)
        bfree = max(0, (info.get("capacity", 0) - info.get("used", 0)) // bsize)
        return abi.KSTATFS.pack(blocks, bfree, bfree,
                                1 << 30, (1 << 30) - info.get("inode_num", 0),
                                bsize, 255, bsize, 0)

    def op_access(self, nodeid, body, ctx):
        return b""

    def op_getxattr(self, nodeid, body, ctx):
        size, _ = abi.GETXATTR_IN.unpack_from(body, 0)
        name = bytes(body[abi.GETXATTR_IN.size:]).split(b"\x00", 1)[0].decode()
        st = self.stat_path(nodeid, self.node_path(nodeid))
        val = st.xattrs.get(name)
        if val is None:
            raise OSError(errno.ENODATA, name)
        val = bytes(val)
        if size == 0:
            return abi.GETXATTR_OUT.pack(len(val), 0)
        if len(val) > size:
            raise OSError(errno.ERANGE, name)
        return val

    def op_listxattr(self, nodeid, body, ctx):
        size, _ = abi.GETXATTR_IN.unpack_from(body, 0)
        st = self.stat_path(nodeid, self.node_path(nodeid))
        blob = b"".join(k.encode() + b"\x00" for k in st.xattrs)
        if size == 0:
            return abi.GETXATTR_OUT.pack(len(blob), 0)
        if len(blob) > size:
            raise OSError(errno.ERANGE, "listxattr")
        return blob

    def op_setxattr(self, nodeid, body, ctx):
        vsize, _flags = abi.SETXATTR_IN.unpack_from(body, 0)
        rest = bytes(body[abi.SETXATTR_IN.size:])
        name, _, tail = rest.partition(b"\x00")
        value = tail[:vsize]
        path = self.node_path(nodeid)
        self.call(self.fs.set_attr(path, xattrs={name.decode(): value}))
        self.invalidate(nodeid)
        return b""

    def op_removexattr(self, nodeid, body, ctx):
        name = bytes(body).split(b"\x00", 1)[0].decode()
        path = self.node_path(nodeid)
        st = self.stat_path(nodeid, path)
        if name not in st.xattrs:
            raise OSError(errno.ENODATA, name)
        self.call(self.fs.set_attr(path, xattrs={name: None}))
        self.invalidate(nodeid)
        return b""

    def op_fallocate(self, nodeid, body, ctx):
        fh, offset, length, mode, _ = abi.FALLOCATE_IN.unpack_from(body, 0)
        FALLOC_FL_KEEP_SIZE = 0x01
        if mode == FALLOC_FL_KEEP_SIZE:
            return b""   # preallocation hint; arenas reserve per block
        if mode != 0:
            raise OSError(errno.ENOTSUP, "fallocate mode")
        # plain fallocate extends the file when offset+length > size
        # (posix_fallocate contract; write_handler.rs resize analog)
        path = self.node_path(nodeid)
        st = self.stat_path(nodeid, path)
        st = self._adjust_writing_size(nodeid, st)
        if offset + length > st.length:
            self.invalidate(nodeid)
            self._truncate(nodeid, path, offset + length, fh)
        return b""

    def op_lseek(self, nodeid, body, ctx):
        fh, offset, whence, _ = abi.LSEEK_IN.unpack_from(body, 0)
        h = self.get_handle(fh)
        length = h.status.length if h.status else 0
        SEEK_DATA, SEEK_HOLE = 3, 4
        if whence not in (SEEK_DATA, SEEK_HOLE):
            raise OSError(errno.EINVAL, "whence")
        if offset >= length:
            raise OSError(errno.ENXIO, "past eof")
        # consult the open reader's extent map: ranges the cached blocks
        # do not cover are holes (sparse files, extending truncate)
        blocks = getattr(getattr(h, "reader", None), "fb", None)
        blocks = blocks.blocks if blocks is not None else []
        if whence == SEEK_DATA:
            for lb in blocks:
                end = lb.offset + lb.block.length
                if offset < end:
                    return abi.LSEEK_OUT.pack(max(offset, lb.offset))
            raise OSError(errno.ENXIO, "in trailing hole")
        # SEEK_HOLE: first uncovered byte at or after offset
        pos = offset
        for lb in blocks:
            end = lb.offset + lb.block.length
            if pos < lb.offset:
                return abi.LSEEK_OUT.pack(pos)     # gap hole
            if pos < end:
                pos = end
        return abi.LSEEK_OUT.pack(min(pos, length)
                                  if pos < length else length)

    def op_interrupt(self, nodeid, body, ctx):
        """INTERRUPT carries the unique of an in-flight request; blocked
        SETLKW waiters poll self.interrupted and abort with EINTR."""
        import struct as _st
        (unique,) = _st.unpack_from("<Q", body, 0)
        with self.plock_mu:
            self.interrupted.add(unique)
            if len(self.interrupted) > 4096:
                self.interrupted.clear()   # stale uniques: reset
        return None

    # ---------------- POSIX advisory locks ----------------
    F_RDLCK, F_WRLCK, F_UNLCK = 0, 1, 2

    def _lock_conflicts(self, node_id, start, end, ltype, owner):
        for (s, e, t, o, _pid) in self.plocks.get(node_id, []):
            if o == owner:
                continue
            if s <= end and start <= e and (t == self.F_WRLCK or
                                            ltype == self.F_WRLCK):
                return (s, e, t, o)
        return None

    def op_getlk(self, nodeid, body, ctx):
        fh, owner, start, end, ltype, pid, _fl, _ = abi.LK_IN.unpack_from(body, 0)
        with self.plock_mu:
            c = self._lock_conflicts(nodeid, start, end or (1 << 63), ltype, owner)
        if c is None:
            return abi.LK_OUT.pack(0, 0, self.F_UNLCK, 0)
        return abi.LK_OUT.pack(c[0], c[1], c[2], 0)

    def op_setlk(self, nodeid, body, ctx, wait=False):
        fh, owner, start, end, ltype, pid, _fl, _ = abi.LK_IN.unpack_from(body, 0)
        end = end or (1 << 63)
        unique = ctx[3] if len(ctx) > 3 else 0
        deadline = time.monotonic() + 300 if wait else 0
        while True:
            with self.plock_mu:
                if ltype == self.F_UNLCK:
                    locks = self.plocks.get(nodeid, [])
                    self.plocks[nodeid] = [
                        l for l in locks
                        if not (l[3] == owner and l[0] <= end and start <= l[1])]
                    return b""
                c = self._lock_conflicts(nodeid, start, end, ltype, owner)
                if c is None:
                    self.plocks.setdefault(nodeid, []).append(
                        (start, end, ltype, owner, pid))
                    return b""
                if unique and unique in self.interrupted:
                    # the kernel INTERRUPTed this request (signal on the
                    # blocked thread): abort the wait
                    self.interrupted.discard(unique)
                    raise OSError(errno.EINTR, "lock wait interrupted")
            if not wait or time.monotonic() > deadline:
                raise OSError(errno.EAGAIN, "lock conflict")
            time.sleep(0.02)

    def op_setlkw(self, nodeid, body, ctx):
        """Blocking lock waits run on their own thread (reply deferred):
        a wait parked on the channel thread would wedge that channel —
        including the INTERRUPT that is supposed to abort the wait."""
        ch = ctx[4] if len(ctx) > 4 else None
        unique = ctx[3] if len(ctx) > 3 else 0
        if ch is None:
            return self.op_setlk(nodeid, body, ctx, wait=True)
        body_copy = bytes(body)

        def waiter():
            try:
                res = self.op_setlk(nodeid, body_copy, ctx, wait=True)
                ch.reply(unique, res)
            except OSError as e:
                ch.reply_error(unique, e.errno or errno.EIO)
            except Exception:  # noqa: BLE001
                ch.reply_error(unique, errno.EIO)

        threading.Thread(target=waiter, daemon=True,
                         name="fuse-setlkw").start()
        return None

    # ---------------- hot-upgrade state persist/restore ----------------
    # (fs/file_system.rs:219-223 + state/backend_handle.rs:288-310 +
    #  fs-api StateFile analog)
    def dump_state(self) -> dict:
        """Serialize kernel-visible state (node ids, open handles, locks).
        Writers are completed before dumping (freeze)."""
        with self.handles_lock:
            for h in list(self.handles.values()):
                if h.writer is not None:
                    with h.lock:
                        self._sync_native_write(h)
                        st = h.writer.complete()
                        h.writer = None
                        h.status = st
            handles = [{"fh": h.fh, "node_id": h.node_id, "path": h.path,
                        "flags": h.flags, "write_pos": h.write_pos,
                        "is_dir": h.dir_entries is not None or
                        (h.status.is_dir if h.status else False)}
                       for h in self.handles.values()]
        with self.nodes_lock:
            nodes = [{"id": n.id, "parent": n.parent, "name": n.name,
                      "nlookup": n.nlookup}
                     for n in self.nodes.values() if n.id != 1]
        with self.plock_mu:
            plocks = {str(k): v for k, v in self.plocks.items()}
        return {"nodes": nodes, "next_node": self.next_node,
                "handles": handles, "next_fh": self.next_fh,
                "plocks": plocks}

    def restore_state(self, state: dict) -> None:
        with self.nodes_lock:
            for nd in state.get("nodes", []):
                node = Node(nd["id"], nd["parent"], nd["name"])
                node.nlookup = nd["nlookup"]
                self.nodes[node.id] = node
            for node in self.nodes.values():
                parent = self.nodes.get(node.parent)
                if parent is not None and node.id != 1:
                    parent.children[node.name] = node.id
            self.next_node = state.get("next_node", self.next_node)
        with self.handles_lock:
            for hd in state.get("handles", []):
                h = FileHandle(hd["fh"], hd["node_id"], hd["path"])
                h.flags = hd.get("flags", 0)
                h.write_pos = hd.get("write_pos", 0)
                # readers rebuilt lazily on first READ
                self.handles[h.fh] = h
            self.next_fh = state.get("next_fh", self.next_fh)
        with self.plock_mu:
            for k, v in state.get("plocks", {}).items():
                self.plocks[int(k)] = [tuple(x) for x in v]

    HANDLERS = {}


CurvineFuseFs.HANDLERS = {
    abi.Op.INIT: CurvineFuseFs.op_init,
    abi.Op.DESTROY: CurvineFuseFs.op_destroy,
    abi.Op.LOOKUP: CurvineFuseFs.op_lookup,
    abi.Op.FORGET: CurvineFuseFs.op_forget,
    abi.Op.BATCH_FORGET: CurvineFuseFs.op_batch_forget,
    abi.Op.GETATTR: CurvineFuseFs.op_getattr,
    abi.Op.SETATTR: CurvineFuseFs.op_setattr,
    abi.Op.MKDIR: CurvineFuseFs.op_mkdir,
    abi.Op.UNLINK: CurvineFuseFs.op_unlink,
    abi.Op.RMDIR: CurvineFuseFs.op_rmdir,
    abi.Op.RENAME: CurvineFuseFs.op_rename,
    abi.Op.RENAME2: CurvineFuseFs.op_rename2,
    abi.Op.SYMLINK: CurvineFuseFs.op_symlink,
    abi.Op.READLINK: CurvineFuseFs.op_readlink,
    abi.Op.LINK: CurvineFuseFs.op_link,
    abi.Op.MKNOD: CurvineFuseFs.op_mknod,
    abi.Op.CREATE: CurvineFuseFs.op_create,
    abi.Op.OPEN: CurvineFuseFs.op_open,
    abi.Op.READ: CurvineFuseFs.op_read,
    abi.Op.WRITE: CurvineFuseFs.op_write,
    abi.Op.FLUSH: CurvineFuseFs.op_flush,
    abi.Op.FSYNC: CurvineFuseFs.op_fsync,
    abi.Op.RELEASE: CurvineFuseFs.op_release,
    abi.Op.OPENDIR: CurvineFuseFs.op_opendir,
    abi.Op.READDIR: CurvineFuseFs.op_readdir,
    abi.Op.READDIRPLUS: CurvineFuseFs.op_readdirplus,
    abi.Op.RELEASEDIR: CurvineFuseFs.op_releasedir,
    abi.Op.FSYNCDIR: CurvineFuseFs.op_fsyncdir,
    abi.Op.STATFS: CurvineFuseFs.op_statfs,
    abi.Op.ACCESS: CurvineFuseFs.op_access,
    abi.Op.GETXATTR: CurvineFuseFs.op_getxattr,
    abi.Op.LISTXATTR: CurvineFuseFs.op_listxattr,
    abi.Op.SETXATTR: CurvineFuseFs.op_setxattr,
    abi.Op.REMOVEXATTR: CurvineFuseFs.op_removexattr,
    abi.Op.FALLOCATE: CurvineFuseFs.op_fallocate,
    abi.Op.LSEEK: CurvineFuseFs.op_lseek,
    abi.Op.INTERRUPT: CurvineFuseFs.op_interrupt,
    abi.Op.GETLK: CurvineFuseFs.op_getlk,
    abi.Op.SETLK: CurvineFuseFs.op_setlk,
    abi.Op.SETLKW: CurvineFuseFs.op_setlkw,
}
