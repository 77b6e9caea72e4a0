"""Native worker data-plane glue.

Pairs with csrc/data_server.cpp: C++ epoll threads own the worker's
listening socket, serve ReadBlock streams straight from arenas/files and
consume WriteBlock data frames into reserved extents; stream control
frames (Open/Complete/Cancel), heartbeats and everything else are
forwarded here to the ordinary Python ``WorkerHandler``.

This is the MI355X answer to the reference's splice/sendfile worker hot
path (crates/core/rpc/src/handler/rpc_frame.rs:82-148,
curvine-worker/src/worker/handler/read_handler.rs:183-214): remote block
reads and replication pushes never cross the interpreter per chunk.
"""
from __future__ import annotations

import asyncio
import logging
import socket
from typing import Optional

from curvine_amd import native
from curvine_amd.rpc.message import Message, PROTO_SIZE

log = logging.getLogger("curvine.worker.native")


class DataPlane:
    """BlockStore observer: mirrors finalized blocks into the C++ read
    registry (publish/drop) and exposes native reader refcounts so
    deletes can defer extent frees."""

    def __init__(self, lib, sid: int):
        self.lib = lib
        self.sid = sid

    def publish(self, block_id: int, layout, meta: dict) -> None:
        try:
            if meta.get("kind") == "arena":
                self.lib.data_block_publish(
                    self.sid, block_id, 0, layout.arena.handle,
                    meta["offset"], meta["length"], "")
            else:
                self.lib.data_block_publish(
                    self.sid, block_id, 1, -1, 0, meta["length"],
                    meta["path"], getattr(layout, "direct", False))
        except Exception as e:  # noqa: BLE001 — registry is a cache
            log.warning("publish block %d: %s", block_id, e)

    def drop(self, block_id: int) -> int:
        try:
            return self.lib.data_block_drop(self.sid, block_id)
        except Exception:  # noqa: BLE001
            return 0

    def refs(self, block_id: int) -> int:
        try:
            return self.lib.data_block_refs(self.sid, block_id)
        except Exception:  # noqa: BLE001
            return 0


class _FwdConn:
    """Stands in for rpc.server.ServerConn on forwarded frames; streaming
    handlers (the Python _read_block fallback) send through the native
    connection."""

    def __init__(self, frontend: "NativeDataFrontend", conn_id: int):
        self.frontend = frontend
        self.conn_id = conn_id
        self.peer = f"native:{conn_id}"
        self.state: dict = {}

    async def send(self, msg: Message) -> None:
        enc = msg.encode()
        loop = asyncio.get_event_loop()
        ok = await loop.run_in_executor(
            None, self.frontend.lib.data_send, self.frontend.sid,
            self.conn_id, enc)
        if not ok:
            raise ConnectionResetError("native data conn gone")


class NativeDataFrontend:
    """Owns the listening socket + C++ data server; drop-in for the
    worker's RpcServer slot (same .port / .start() / .stop() surface)."""

    def __init__(self, worker, nthreads: int = 4, fwd_batch: int = 64):
        self.worker = worker
        self.lib = native.load()
        if not hasattr(self.lib, "data_create"):
            raise RuntimeError("_native.so lacks data server (rebuild)")
        conf = worker.conf.worker
        sock = socket.socket(socket.AF_INET, socket.SOCK_STREAM)
        sock.setsockopt(socket.SOL_SOCKET, socket.SO_REUSEADDR, 1)
        sock.bind((conf.hostname, conf.rpc_port))
        sock.listen(1024)
        self.port = sock.getsockname()[1]
        self.sid = self.lib.data_create(sock.detach(), nthreads)
        self.data_plane = DataPlane(self.lib, self.sid)
        self.fwd_batch = fwd_batch
        self._loop: Optional[asyncio.AbstractEventLoop] = None
        self._efd: Optional[int] = None
        self._stopped = False
        # per-connection serial drain (pipelined frames execute in order)
        self._queues: dict[int, asyncio.Queue] = {}
        self._tasks: dict[int, asyncio.Task] = {}
        self._handlers: dict[int, object] = {}

    # ---------------- lifecycle ----------------
    async def start(self) -> None:
        self._loop = asyncio.get_running_loop()
        store = self.worker.store
        store.data_plane = self.data_plane
        # publish everything already finalized (startup scan / restarts)
        from curvine_amd.worker.block_store import BlockState
        with store.lock:
            snapshot = [(b.block_id, b.layout, b.meta)
                        for b in store.blocks.values()
                        if b.state == BlockState.FINALIZED
                        and not b.pending_delete]
        for bid, layout, meta in snapshot:
            self.data_plane.publish(bid, layout, meta)
        self._efd = self.lib.data_eventfd(self.sid)
        self._loop.add_reader(self._efd, self._on_forward_ready)
        log.info("native data frontend on :%d (%d blocks published)",
                 self.port, len(snapshot))

    async def stop(self) -> None:
        self._stopped = True
        if self._loop is not None and self._efd is not None:
            self._loop.remove_reader(self._efd)
        self.lib.data_stop(self.sid)
        for t in self._tasks.values():
            t.cancel()
        if self._tasks:
            await asyncio.gather(*self._tasks.values(),
                                 return_exceptions=True)
        self._tasks.clear()
        self._queues.clear()
        self._handlers.clear()

    def stats(self) -> dict:
        return self.lib.data_stats(self.sid)

    # ---------------- forwarded frames ----------------
    def _on_forward_ready(self) -> None:
        import os
        try:
            os.read(self._efd, 8)
        except BlockingIOError:
            pass
        except OSError:
            return
        while True:
            items = self.lib.data_forward_pop(self.sid, 0, self.fwd_batch)
            if not items:
                break
            for conn_id, raw in items:
                if not raw:   # close sentinel
                    t = self._tasks.pop(conn_id, None)
                    if t is not None:
                        t.cancel()
                    self._queues.pop(conn_id, None)
                    h = self._handlers.pop(conn_id, None)
                    if h is not None:
                        asyncio.ensure_future(h.on_close())
                    continue
                q = self._queues.get(conn_id)
                if q is None:
                    q = self._queues[conn_id] = asyncio.Queue()
                    self._tasks[conn_id] = asyncio.ensure_future(
                        self._drain(conn_id, q))
                q.put_nowait(raw)

    def _handler_for(self, conn_id: int):
        h = self._handlers.get(conn_id)
        if h is None:
            from curvine_amd.worker.native_data import NativeWorkerHandler
            h = NativeWorkerHandler(self.worker, self, conn_id)
            self._handlers[conn_id] = h
        return h

    async def _drain(self, conn_id: int, q: asyncio.Queue) -> None:
        conn = _FwdConn(self, conn_id)
        handler = self._handler_for(conn_id)
        try:
            while True:
                raw = await q.get()
                try:
                    hlen, dlen, msg = Message.decode_proto(raw[:PROTO_SIZE])
                    if hlen:
                        msg.set_header_bytes(raw[PROTO_SIZE:PROTO_SIZE + hlen])
                    if dlen:
                        msg.data = raw[PROTO_SIZE + hlen:
                                       PROTO_SIZE + hlen + dlen]
                    try:
                        reply = await handler.handle(msg, conn)
                    except asyncio.CancelledError:
                        raise
                    except Exception as e:  # noqa: BLE001
                        log.debug("fwd handler error code=%s: %s",
                                  msg.code, e)
                        reply = msg.error_reply(e)
                    if reply is not None:
                        ok = await asyncio.get_event_loop().run_in_executor(
                            None, self.lib.data_send, self.sid, conn_id,
                            reply.encode())
                        if not ok:
                            break
                except asyncio.CancelledError:
                    raise
                except Exception:  # noqa: BLE001
                    log.exception("forwarded data frame dispatch failed")
        except asyncio.CancelledError:
            pass
        finally:
            self._tasks.pop(conn_id, None)
            self._queues.pop(conn_id, None)


class NativeWorkerHandler:
    """WorkerHandler wrapper that registers native write sessions so the
    C++ loop consumes the stream's data frames directly."""

    def __init__(self, worker, frontend: NativeDataFrontend, conn_id: int):
        from curvine_amd.worker.handlers import WorkerHandler
        self.inner = WorkerHandler(worker)
        self.frontend = frontend
        self.conn_id = conn_id

    async def on_close(self):
        await self.inner.on_close()

    async def handle(self, msg: Message, conn) -> Optional[Message]:
        from curvine_amd.rpc.codes import RpcCode
        from curvine_amd.rpc.message import Status
        if msg.code != int(RpcCode.WriteBlock):
            return await self.inner.handle(msg, conn)
        lib, sid = self.frontend.lib, self.frontend.sid
        if msg.req_status == Status.Open:
            reply = await self.inner.handle(msg, conn)
            sess = self.inner.writes.get(msg.req_id)
            if sess is not None:
                w = sess["writer"]
                meta = w.meta
                try:
                    if meta.get("kind") == "arena":
                        lib.data_write_register(
                            sid, self.conn_id, msg.req_id, 0,
                            w.layout.arena.handle, meta["offset"],
                            meta["reserved"], -1, w.pos)
                    elif meta.get("_f") is not None:
                        lib.data_write_register(
                            sid, self.conn_id, msg.req_id, 1, -1, 0,
                            meta["reserved"], meta["_f"].fileno(), w.pos)
                except Exception as e:  # noqa: BLE001 — Python fallback
                    log.debug("native write session register failed: %s", e)
            return reply
        if msg.req_status in (Status.Complete, Status.Cancel):
            pos = lib.data_write_unregister(sid, self.conn_id, msg.req_id)
            sess = self.inner.writes.get(msg.req_id)
            if pos >= 0 and sess is not None:
                # natively-consumed bytes advance the finalize watermark
                w = sess["writer"]
                w.pos = max(w.pos, pos)
            return await self.inner.handle(msg, conn)
        # Running frames only reach Python when the native session was not
        # registered (fallback) or the payload overflowed the reservation
        return await self.inner.handle(msg, conn)
