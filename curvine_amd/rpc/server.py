"""Asyncio TCP RPC server.

Analog of the reference's `RpcServer`
(/root/reference/crates/core/rpc/src/server/rpc_server.rs:27-232): an accept
loop spawning one task per connection, with a per-connection stateful
handler supplied by a `HandlerService`
(handler/handler_service.rs:46-76), plus connection/global concurrency
limits (LimitConf analog).
"""
from __future__ import annotations

import asyncio
import logging
import socket
from typing import Optional

from curvine_amd.rpc.message import Message, PROTO_SIZE, MAX_DATA_SIZE

log = logging.getLogger("curvine.rpc")


class ServerConn:
    """Server side of one TCP connection; passed to handlers for streaming
    replies."""

    def __init__(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        self.reader = reader
        self.writer = writer
        self._wlock = asyncio.Lock()
        peer = writer.get_extra_info("peername")
        self.peer = f"{peer[0]}:{peer[1]}" if peer else "?"
        self.state: dict = {}   # per-connection handler scratch

    async def send(self, msg: Message) -> None:
        parts = msg.encode_parts()
        async with self._wlock:
            self.writer.write(parts[0])
            if parts[1]:
                self.writer.write(parts[1])
            # drain (an extra task switch) only when backpressure matters:
            # large payloads, or the socket buffer has actually filled up
            if len(parts[1]) >= (64 << 10) or \
                    self.writer.transport.get_write_buffer_size() > (1 << 20):
                await self.writer.drain()

    # buffered frame parsing: pipelined requests arrive in one TCP read, so
    # parse as many frames per await as the buffer holds (2-3 awaits per
    # request otherwise dominate metadata QPS)
    _buf: bytearray

    async def recv(self) -> Optional[Message]:
        buf = getattr(self, "_buf", None)
        if buf is None:
            buf = self._buf = bytearray()
        while True:
            if len(buf) >= PROTO_SIZE:
                hlen, dlen, msg = Message.decode_proto(bytes(buf[:PROTO_SIZE]))
                if dlen > MAX_DATA_SIZE:
                    raise ValueError(
                        f"frame data_len {dlen} exceeds {MAX_DATA_SIZE}")
                if hlen > MAX_DATA_SIZE:
                    # corrupt/malicious frame: never buffer an unbounded
                    # header
                    raise ValueError(
                        f"frame header_len {hlen} exceeds {MAX_DATA_SIZE}")
                total = PROTO_SIZE + hlen + dlen
                if len(buf) >= total:
                    if hlen:
                        msg.set_header_bytes(
                            bytes(buf[PROTO_SIZE:PROTO_SIZE + hlen]))
                    if dlen:
                        msg.data = bytes(buf[PROTO_SIZE + hlen:total])
                    del buf[:total]
                    return msg
            try:
                chunk = await self.reader.read(256 << 10)
            except ConnectionResetError:
                return None
            if not chunk:
                return None
            buf += chunk


class HandlerService:
    """Per-connection handler factory. Subclass and override
    `get_message_handler`; the returned object's `handle(msg, conn)`
    coroutine is called for every inbound frame. Returning a Message sends
    it; returning None means the handler sent (or will send) replies
    itself."""

    def get_message_handler(self):
        raise NotImplementedError


class RpcServer:
    def __init__(self, name: str, hostname: str, port: int,
                 service: HandlerService, max_conns: int = 10_000):
        self.name = name
        self.hostname = hostname
        self.port = port
        self.service = service
        self._server: Optional[asyncio.AbstractServer] = None
        self._conn_sem = asyncio.Semaphore(max_conns)
        self._conns: set[asyncio.Task] = set()

    async def start(self) -> None:
        self._server = await asyncio.start_server(
            self._on_conn, self.hostname, self.port,
            reuse_address=True, limit=MAX_DATA_SIZE + (1 << 20))
        if self.port == 0:   # test support: ephemeral port
            self.port = self._server.sockets[0].getsockname()[1]
        log.info("%s rpc server listening on %s:%d", self.name, self.hostname, self.port)

    async def stop(self) -> None:
        if self._server:
            self._server.close()
            await self._server.wait_closed()
        for t in list(self._conns):
            t.cancel()
        if self._conns:
            await asyncio.gather(*self._conns, return_exceptions=True)

    async def _on_conn(self, reader: asyncio.StreamReader, writer: asyncio.StreamWriter):
        sock = writer.get_extra_info("socket")
        if sock is not None:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        conn = ServerConn(reader, writer)
        handler = self.service.get_message_handler()
        task = asyncio.current_task()
        if task:
            self._conns.add(task)
        try:
            async with self._conn_sem:
                while True:
                    msg = await conn.recv()
                    if msg is None:
                        break
                    if msg.code == 1 and not msg.header and not msg.data:
                        # lightweight keepalive (rpc_frame.rs:278-283 analog)
                        await conn.send(msg.reply())
                        continue
                    try:
                        reply = await handler.handle(msg, conn)
                    except asyncio.CancelledError:
                        raise
                    except Exception as e:  # noqa: BLE001 — errors cross the wire
                        log.debug("%s handler error code=%s: %s", self.name, msg.code, e)
                        reply = msg.error_reply(e)
                    if reply is not None:
                        await conn.send(reply)
        except asyncio.CancelledError:
            pass
        except Exception as e:  # noqa: BLE001
            log.debug("%s conn %s closed: %s", self.name, conn.peer, e)
        finally:
            close = getattr(handler, "on_close", None)
            if close is not None:
                try:
                    r = close()
                    if asyncio.iscoroutine(r):
                        await r
                except Exception:  # noqa: BLE001
                    pass
            if task:
                self._conns.discard(task)
            try:
                writer.close()
                await writer.wait_closed()
            except Exception:  # noqa: BLE001
                pass
