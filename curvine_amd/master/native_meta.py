"""Native metadata RPC frontend glue.

Pairs with csrc/meta_server.cpp: the C++ epoll threads own the master's
listening socket and serve FileStatus/ListStatus/Exists (and keepalives)
from a GIL-free mirror of the inode tree; every other frame is forwarded
here and handled by the ordinary Python MasterHandler, the reply written
back through ``meta_send``.

This is the MI355X answer to the reference's tokio-native RPC server
(crates/core/rpc/src/server/rpc_server.rs): the metadata read path —
the QPS benchmark surface (curvine-tests fs bench) — never crosses the
interpreter.

Mirror maintenance: ``MetaMirror`` hangs off ``FsDir.mirror``; every
``_apply_*`` (the shared live/replay choke point, fs_dir.py) notifies it
synchronously, so a mutation is visible in the C++ tree before the
mutating RPC's reply leaves the master.  Whole-state swaps (snapshot
install, step-down rebuild) call ``attach()`` to re-prime the mirror.
"""
from __future__ import annotations

import asyncio
import logging
import socket
import struct
from typing import Optional

import msgpack

from curvine_amd import native
from curvine_amd.conf import TIER_ORDER
from curvine_amd.model import WorkerState, now_ms
from curvine_amd.rpc.message import Message, PROTO_SIZE

log = logging.getLogger("curvine.meta.native")


def _pack_pairs(d: dict) -> tuple[bytes, int]:
    """msgpack-encode a dict, returning (pairs-without-map-header, npairs)
    so C++ can splice extra keys (``path``) into the same map."""
    b = msgpack.packb(d, use_bin_type=True)
    first = b[0]
    if 0x80 <= first <= 0x8F:
        return b[1:], first & 0xF
    if first == 0xDE:
        return b[3:], int.from_bytes(b[1:3], "big")
    if first == 0xDF:
        return b[5:], int.from_bytes(b[1:5], "big")
    raise ValueError("not a msgpack map")


def _node_blob(node) -> tuple[bytes, int]:
    """FileStatus fields (fs_dir.status_of / model.FileStatus.to_dict)
    minus the lookup-dependent ``path`` and ``mtime_ms`` (kept out of the
    blob so parent-mtime touches never repack — passed separately)."""
    return _pack_pairs({
        "inode_id": node.id, "name": node.name,
        "file_type": int(node.file_type), "length": node.length,
        "is_complete": node.complete, "block_size": node.block_size,
        "replicas": node.replicas, "storage_tier": node.storage_tier,
        "atime_ms": node.atime_ms,
        "mode": node.mode, "uid": node.uid, "gid": node.gid,
        "ttl_ms": node.ttl_ms, "ttl_action": node.ttl_action,
        "symlink_target": node.symlink_target, "nlink": node.nlink,
        "xattrs": {k: bytes(v) for k, v in node.xattrs.items()},
    })


def _pack_blocks(node) -> bytes:
    if not node.blocks:
        return b""
    flat = []
    for bid, blen in node.blocks:
        flat.append(bid)
        flat.append(blen)
    return struct.pack(f"<{len(flat)}q", *flat)


class MetaMirror:
    """FsDir observer pushing inode state into the C++ tree."""

    def __init__(self, lib, sid: int):
        self.lib = lib
        self.sid = sid

    def upsert(self, node) -> None:
        # positional fast path: the blob is packed in C++ (no Python dict
        # + msgpack per mutation — the mutation-QPS hot path)
        xb = None
        if node.xattrs:
            import msgpack as _mp
            xb = _mp.packb({k: bytes(v) for k, v in node.xattrs.items()},
                           use_bin_type=True)
        self.lib.meta_upsert_node(
            self.sid, node.id, node.is_dir, node.name, int(node.file_type),
            node.length, node.complete, node.block_size, node.replicas,
            node.storage_tier, node.mtime_ms, node.atime_ms, node.mode,
            node.uid, node.gid, node.ttl_ms, node.ttl_action,
            node.symlink_target, node.nlink, _pack_blocks(node), xb)

    def touch(self, inode_id: int, mtime_ms: int) -> None:
        self.lib.meta_touch(self.sid, inode_id, mtime_ms)

    def _raw_upsert(self, inode_id: int, is_dir: bool, blob: bytes,
                    n: int, blocks: bytes, mtime: int) -> None:
        """Pre-serialized upsert (CommitGatedMirror replays these)."""
        self.lib.meta_upsert(self.sid, inode_id, is_dir, blob, n, blocks,
                             mtime)

    def add_child(self, parent_id: int, name: str, child_id: int) -> None:
        self.lib.meta_add_child(self.sid, parent_id, name, child_id)

    def remove_child(self, parent_id: int, name: str) -> None:
        self.lib.meta_remove_child(self.sid, parent_id, name)

    def drop(self, inode_id: int) -> None:
        self.lib.meta_drop(self.sid, inode_id)


class CommitGatedMirror:
    """Linearizable native reads under raft: the leader applies entries
    to FsDir at APPEND time (optimistic), but the C++ tree must only see
    COMMITTED state — a concurrent native FileStatus would otherwise
    observe data a step-down rebuild later erases.

    Each mirror op is serialized eagerly (capturing the node's state at
    apply time, before any later mutation touches it), tagged with the
    entry's op_id (== raft index), and replayed into the C++ tree when
    the commit index passes it.  Follower applies are always committed
    already, so they pass straight through."""

    def __init__(self, inner: MetaMirror, current_op, current_commit):
        self.inner = inner
        self._op = current_op            # () -> op_id being applied
        self._commit = current_commit    # () -> raft commit index
        self.pending: list[tuple[int, tuple]] = []   # (op_id, call)

    def _gate(self, call: tuple) -> None:
        op = self._op()
        if op <= self._commit():
            self._run(call)
        else:
            self.pending.append((op, call))

    def _run(self, call: tuple) -> None:
        fn, *args = call
        getattr(self.inner, fn)(*args)

    def flush(self, commit: int) -> None:
        if not self.pending:
            return
        i = 0
        for i, (op, call) in enumerate(self.pending):  # noqa: B007
            if op > commit:
                break
            self._run(call)
        else:
            i += 1
        del self.pending[:i]

    # MetaMirror surface — serialize NOW, apply at commit
    def upsert(self, node) -> None:
        blob, n = _node_blob(node)
        self._gate(("_raw_upsert", node.id, node.is_dir, blob, n,
                    _pack_blocks(node), node.mtime_ms))

    def touch(self, inode_id: int, mtime_ms: int) -> None:
        self._gate(("touch", inode_id, mtime_ms))

    def add_child(self, parent_id: int, name: str, child_id: int) -> None:
        self._gate(("add_child", parent_id, name, child_id))

    def remove_child(self, parent_id: int, name: str) -> None:
        self._gate(("remove_child", parent_id, name))

    def drop(self, inode_id: int) -> None:
        self._gate(("drop", inode_id))


class WorkerMirror:
    """WorkerManager observer: worker addresses + block locations, so the
    C++ frontend can assemble OpenFile/GetBlockLocations replies."""

    def __init__(self, lib, sid: int):
        self.lib = lib
        self.sid = sid

    def upsert_worker(self, info) -> None:
        blob = msgpack.packb(info.address.to_dict(), use_bin_type=True)
        self.lib.meta_worker_upsert(self.sid, info.address.worker_id, blob,
                                    int(info.state) == int(WorkerState.LOST))

    def add_loc(self, block_id: int, worker_id: int, tier: str) -> None:
        self.lib.meta_block_add_loc(self.sid, block_id, worker_id, tier,
                                    TIER_ORDER.get(tier, 9))

    def remove_loc(self, block_id: int, worker_id: int) -> None:
        self.lib.meta_block_remove_loc(self.sid, block_id, worker_id)

    def drop_block(self, block_id: int) -> None:
        self.lib.meta_block_drop(self.sid, block_id)


class _Delegate:
    """Manual ``yield from`` for a coroutine that already yielded once:
    re-yields its pending awaitable chain so a Task can finish driving it."""

    def __init__(self, coro, first_yield):
        self.coro = coro
        self.first = first_yield

    def __await__(self):
        coro, y = self.coro, self.first
        try:
            while True:
                sent = yield y
                y = coro.send(sent)
        except StopIteration as si:
            return si.value


class _BadFrame(Exception):
    """Frame whose msgpack header failed to decode: reply with a wire
    error instead of letting the exception escape the dispatch loop."""

    def __init__(self, msg, cause):
        super().__init__(str(cause))
        self.msg = msg
        self.cause = cause if isinstance(cause, Exception) else Exception(cause)


class _FwdConn:
    """Shim standing in for rpc.server.ServerConn on forwarded frames
    (master handlers are unary and never touch it beyond attributes)."""
    __slots__ = ("peer", "state")

    def __init__(self, conn_id: int):
        self.peer = f"native:{conn_id}"
        self.state = {}


class NativeMetaFrontend:
    """Owns the listening socket + C++ server; drop-in for the master's
    RpcServer slot (same .port / .start() / .stop() surface)."""

    def __init__(self, master, nthreads: int = 4, fwd_batch: int = 256):
        self.master = master
        self.lib = native.load()
        if not hasattr(self.lib, "meta_create"):
            raise RuntimeError("_native.so lacks meta server (rebuild)")
        conf = master.conf.master
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind((conf.hostname, conf.rpc_port))
        sock.listen(1024)
        self.port = sock.getsockname()[1]
        self.sid = self.lib.meta_create(sock.detach(), nthreads)
        self.mirror_gate: Optional[CommitGatedMirror] = None
        self.fwd_batch = fwd_batch
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._efd: Optional[int] = None
        self._handler = None
        self._stopped = False
        import os
        self._inline = os.environ.get("CURVINE_META_INLINE", "1") != "0"
        # per-connection serial drain: pipelined requests on one socket
        # must execute in arrival order (the asyncio server's per-conn
        # loop gave that for free)
        self._queues: dict[int, asyncio.Queue] = {}
        self._tasks: dict[int, asyncio.Task] = {}

    # ---------------- mirror ----------------
    def attach(self) -> None:
        """(Re)prime the C++ tree from the current FsDir + WorkerManager
        and hook future mutations.  Called at start and after
        snapshot-install/rebuild."""
        from curvine_amd.master.fs_dir import MirrorFanout
        fs_dir = self.master.fs.fs_dir
        mirror = MetaMirror(self.lib, self.sid)
        raft = self.master.raft
        if raft is not None:
            # attach() runs from a clean (all-committed) state: rebuild,
            # snapshot install, and start all discard any optimistic tail
            # first, so an empty pending queue is correct here
            mirror = self.mirror_gate = CommitGatedMirror(
                mirror, lambda: self.master.journal.op_id,
                lambda: raft.commit_index)
            raft.on_commit = self.mirror_gate.flush
        # preserve co-observers (e.g. the sqlite inode store)
        cur = fs_dir.mirror
        others = []
        _native = (MetaMirror, CommitGatedMirror)
        if isinstance(cur, MirrorFanout):
            others = [m for m in cur.mirrors if not isinstance(m, _native)]
        elif cur is not None and not isinstance(cur, _native):
            others = [cur]
        fs_dir.mirror = MirrorFanout([mirror] + others) if others else mirror
        self.lib.meta_clear(self.sid)
        up, ac = self.lib.meta_upsert, self.lib.meta_add_child
        for node in fs_dir.inodes.values():
            blob, n = _node_blob(node)
            up(self.sid, node.id, node.is_dir, blob, n, _pack_blocks(node),
               node.mtime_ms)
            if node.children:
                for name, cid in node.children.items():
                    ac(self.sid, node.id, name, cid)
        workers = self.master.fs.workers
        wmirror = WorkerMirror(self.lib, self.sid)
        workers.mirror = wmirror
        for info in workers.workers.values():
            wmirror.upsert_worker(info)
        for bid, locs in workers.block_locs.items():
            for wid, tier in locs.items():
                wmirror.add_loc(bid, wid, tier)

    def set_serving(self, on: bool) -> None:
        self.lib.meta_set_serving(self.sid, on)

    def drain_access(self) -> None:
        """Fold native open() access counters back into the Python inodes
        (LFU/LRU eviction inputs; atime approximated to drain time)."""
        counts = self.lib.meta_take_access(self.sid)
        if not counts:
            return
        fs_dir = self.master.fs.fs_dir
        t = now_ms()
        for inode_id, n in counts.items():
            node = fs_dir.inodes.get(inode_id)
            if node is not None:
                node.access_count += n
                node.atime_ms = t

    def stats(self) -> dict:
        return self.lib.meta_stats(self.sid)

    # ---------------- lifecycle ----------------
    async def start(self) -> None:
        self._loop = asyncio.get_running_loop()
        self._handler = self.master.rpc_service.get_message_handler()
        self.attach()
        # forwarded frames wake the loop through an eventfd the C++ side
        # signals on empty->non-empty — the same epoll wake path a socket
        # read gives the asyncio server, with no relay thread in between
        self._efd = self.lib.meta_eventfd(self.sid)
        self._loop.add_reader(self._efd, self._on_forward_ready)
        raft = self.master.raft
        self.set_serving(raft is None or raft.is_leader)
        log.info("native meta frontend on :%d", self.port)

    async def stop(self) -> None:
        self._stopped = True
        if self._loop is not None and self._efd is not None:
            self._loop.remove_reader(self._efd)
        self.lib.meta_stop(self.sid)
        for t in self._tasks.values():
            t.cancel()
        if self._tasks:
            await asyncio.gather(*self._tasks.values(),
                                 return_exceptions=True)
        self._tasks.clear()
        self._queues.clear()

    # ---------------- forwarded frames ----------------
    def _on_forward_ready(self) -> None:
        import os
        try:
            os.read(self._efd, 8)          # clear the signal first
        except BlockingIOError:
            pass
        except OSError:
            return                          # efd closed at stop
        while True:
            items = self.lib.meta_forward_pop(self.sid, 0, self.fwd_batch)
            if not items:
                break
            self._dispatch(items)

    def _dispatch(self, items) -> None:
        # Fast path (no raft): master handlers are synchronous coroutines
        # (the only await, raft.wait_commit, is skipped), so drive each to
        # completion inline with coro.send(None) — no Queue, no Task — and
        # batch every reply for a connection into one meta_send.
        # CURVINE_META_INLINE=0 forces the ordered-queue path (A/B knob).
        sync_ok = self._inline and self.master.raft is None
        replies: dict[int, list[bytes]] = {}
        for conn_id, raw in items:
            if not raw:                      # close sentinel from C++
                t = self._tasks.pop(conn_id, None)
                if t is not None:
                    t.cancel()
                self._queues.pop(conn_id, None)
                continue
            if sync_ok and conn_id not in self._queues:
                enc = self._handle_sync(conn_id, raw)
                if enc is not None:
                    replies.setdefault(conn_id, []).append(enc)
                continue
            q = self._queues.get(conn_id)
            if q is None:
                q = self._queues[conn_id] = asyncio.Queue()
                self._tasks[conn_id] = asyncio.ensure_future(
                    self._drain(conn_id, q))
            q.put_nowait(raw)
        for conn_id, bufs in replies.items():
            self.lib.meta_send(self.sid, conn_id, b"".join(bufs))

    @staticmethod
    def _decode_frame(raw: bytes) -> Message:
        hlen, dlen, msg = Message.decode_proto(raw[:PROTO_SIZE])
        try:
            if hlen:
                msg.set_header_bytes(raw[PROTO_SIZE:PROTO_SIZE + hlen])
        except Exception as e:  # noqa: BLE001 — undecodable header
            raise _BadFrame(msg, e)
        if dlen:
            msg.data = raw[PROTO_SIZE + hlen:PROTO_SIZE + hlen + dlen]
        return msg

    def _handle_sync(self, conn_id: int, raw: bytes) -> Optional[bytes]:
        """Drive one handler coroutine synchronously; falls back to the
        ordered queue path if it unexpectedly suspends."""
        try:
            msg = self._decode_frame(raw)
        except _BadFrame as bf:
            return bf.msg.error_reply(bf.cause).encode()
        coro = self._handler.handle(msg, _FwdConn(conn_id))
        try:
            y = coro.send(None)
        except StopIteration as si:
            reply = si.value
        except Exception as e:  # noqa: BLE001 — errors cross the wire
            log.debug("fwd handler error code=%s: %s", msg.code, e)
            reply = msg.error_reply(e)
        else:
            # suspended (shouldn't happen without raft): finish in a task;
            # route later frames for this conn through the ordered queue
            q = self._queues[conn_id] = asyncio.Queue()
            self._tasks[conn_id] = asyncio.ensure_future(
                self._finish_then_drain(conn_id, q, coro, y, msg))
            return None
        return reply.encode() if reply is not None else None

    async def _finish_then_drain(self, conn_id, q, coro, first_yield, msg):
        try:
            reply = await _Delegate(coro, first_yield)
        except asyncio.CancelledError:
            raise
        except Exception as e:  # noqa: BLE001
            reply = msg.error_reply(e)
        if reply is not None:
            self.lib.meta_send(self.sid, conn_id, reply.encode())
        await self._drain(conn_id, q)

    async def _drain(self, conn_id: int, q: asyncio.Queue) -> None:
        """Serial handler loop for one connection; replies for frames that
        were queued together go out in ONE meta_send."""
        conn = _FwdConn(conn_id)
        buf: list[bytes] = []
        try:
            while True:
                raw = await q.get()
                while True:
                    enc = await self._handle_raw(raw, conn)
                    if enc is not None:
                        buf.append(enc)
                    if q.empty():
                        break
                    raw = q.get_nowait()
                if buf:
                    ok = self.lib.meta_send(self.sid, conn_id, b"".join(buf))
                    buf.clear()
                    if not ok:
                        break
        except asyncio.CancelledError:
            pass
        finally:
            self._tasks.pop(conn_id, None)
            self._queues.pop(conn_id, None)

    async def _handle_raw(self, raw: bytes, conn) -> Optional[bytes]:
        try:
            try:
                msg = self._decode_frame(raw)
            except _BadFrame as bf:
                return bf.msg.error_reply(bf.cause).encode()
            try:
                reply = await self._handler.handle(msg, conn)
            except asyncio.CancelledError:
                raise
            except Exception as e:  # noqa: BLE001 — errors cross the wire
                log.debug("fwd handler error code=%s: %s", msg.code, e)
                reply = msg.error_reply(e)
            return reply.encode() if reply is not None else None
        except asyncio.CancelledError:
            raise
        except Exception:  # noqa: BLE001
            log.exception("forwarded frame dispatch failed")
            return None
