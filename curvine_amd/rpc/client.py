"""Asyncio RPC client, connection pool and master-failover connector.

Analog of the reference client stack
(/root/reference/crates/core/rpc/src/client/): `RpcClient` with a
background demultiplexer matching replies to requests by req_id,
`ClientFactory` pooling connections per address, and `ClusterConnector`
retrying across master addresses on failover (NotLeader / connect errors).
"""
from __future__ import annotations

import asyncio
import logging
import socket
from typing import Optional

from curvine_amd.errors import ConnectError, FsError, NotLeader, RpcTimeout
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.message import Message, Status, PROTO_SIZE, MAX_DATA_SIZE

log = logging.getLogger("curvine.rpc.client")


class RpcClient:
    def __init__(self, hostname: str, port: int, timeout_ms: int = 60_000):
        self.hostname = hostname
        self.port = port
        self.timeout = timeout_ms / 1000.0
        self._reader: Optional[asyncio.StreamReader] = None
        self._writer: Optional[asyncio.StreamWriter] = None
        self._wlock = asyncio.Lock()
        # req_id -> queue of reply frames (queue, because of streams)
        self._pending: dict[int, asyncio.Queue] = {}
        self._rx_task: Optional[asyncio.Task] = None
        self._closed = False

    @property
    def addr(self) -> str:
        return f"{self.hostname}:{self.port}"

    @property
    def is_connected(self) -> bool:
        return self._writer is not None and not self._closed

    async def connect(self) -> "RpcClient":
        try:
            self._reader, self._writer = await asyncio.wait_for(
                asyncio.open_connection(self.hostname, self.port,
                                        limit=MAX_DATA_SIZE + (1 << 20)),
                timeout=self.timeout)
        except (OSError, asyncio.TimeoutError) as e:
            raise ConnectError(f"connect {self.addr}: {e}") from e
        sock = self._writer.get_extra_info("socket")
        if sock is not None:
            sock.setsockopt(socket.IPPROTO_TCP, socket.TCP_NODELAY, 1)
        self._closed = False
        self._rx_task = asyncio.create_task(self._rx_loop())
        return self

    async def close(self) -> None:
        self._closed = True
        if self._rx_task:
            self._rx_task.cancel()
            try:
                await self._rx_task
            except (asyncio.CancelledError, Exception):  # noqa: BLE001
                pass
            self._rx_task = None
        if self._writer:
            try:
                self._writer.close()
                await self._writer.wait_closed()
            except Exception:  # noqa: BLE001
                pass
            self._writer = None
        self._fail_pending(ConnectError(f"{self.addr} closed"))

    def _fail_pending(self, err: Exception) -> None:
        for sink in self._pending.values():
            if isinstance(sink, asyncio.Future):
                if not sink.done():
                    sink.set_exception(err)
            else:
                sink.put_nowait(err)
        self._pending.clear()

    async def _rx_loop(self) -> None:
        buf = bytearray()
        try:
            while True:
                while len(buf) < PROTO_SIZE:
                    chunk = await self._reader.read(256 << 10)
                    if not chunk:
                        raise ConnectionResetError("eof")
                    buf += chunk
                hlen, dlen, msg = Message.decode_proto(bytes(buf[:PROTO_SIZE]))
                if hlen > MAX_DATA_SIZE or dlen > MAX_DATA_SIZE:
                    raise ValueError(
                        f"frame hlen={hlen} dlen={dlen} exceeds "
                        f"{MAX_DATA_SIZE}")
                total = PROTO_SIZE + hlen + dlen
                while len(buf) < total:
                    chunk = await self._reader.read(256 << 10)
                    if not chunk:
                        raise ConnectionResetError("eof")
                    buf += chunk
                if hlen:
                    msg.set_header_bytes(bytes(buf[PROTO_SIZE:PROTO_SIZE + hlen]))
                if dlen:
                    msg.data = bytes(buf[PROTO_SIZE + hlen:total])
                del buf[:total]
                sink = self._pending.get(msg.req_id)
                if sink is None:
                    log.debug("drop orphan reply req_id=%d code=%d",
                              msg.req_id, msg.code)
                elif isinstance(sink, asyncio.Future):
                    if not sink.done():
                        sink.set_result(msg)
                else:
                    sink.put_nowait(msg)
        except asyncio.CancelledError:
            raise
        except Exception as e:  # noqa: BLE001 — propagate to callers
            self._closed = True
            self._fail_pending(ConnectError(f"{self.addr} rx: {e}"))

    async def _send(self, msg: Message) -> None:
        parts = msg.encode_parts()
        async with self._wlock:
            self._writer.write(parts[0])
            if parts[1]:
                self._writer.write(parts[1])
            await self._writer.drain()

    # ---------------- unary ----------------
    async def rpc(self, code: RpcCode, header: dict | None = None,
                  data: bytes = b"", timeout: float | None = None) -> Message:
        if not self.is_connected:
            raise ConnectError(f"{self.addr} not connected")
        msg = Message.request(code, header, data)
        fut = asyncio.get_running_loop().create_future()
        self._pending[msg.req_id] = fut
        try:
            await self._send(msg)
            try:
                reply = await asyncio.wait_for(fut, timeout or self.timeout)
            except asyncio.TimeoutError as e:
                raise RpcTimeout(f"{RpcCode(code).name} to {self.addr} timed out") from e
            return reply.raise_if_error()
        finally:
            self._pending.pop(msg.req_id, None)

    # ---------------- streaming ----------------
    def stream(self, code: RpcCode) -> "RpcStream":
        return RpcStream(self, code)


class RpcStream:
    """Bidirectional stream over one req_id (Open -> Running* -> Complete)."""

    def __init__(self, client: RpcClient, code: RpcCode):
        self.client = client
        self.code = int(code)
        self.req_id = Message.request(code).req_id
        self.seq = 0
        self.q: asyncio.Queue = asyncio.Queue()
        client._pending[self.req_id] = self.q

    async def send(self, header: dict | None = None, data: bytes = b"",
                   status: Status = Status.Running) -> None:
        msg = Message(code=self.code, req_status=status, req_id=self.req_id,
                      seq_id=self.seq, header=header or {}, data=data)
        self.seq += 1
        await self.client._send(msg)

    async def recv(self, timeout: float | None = None) -> Message:
        try:
            reply = await asyncio.wait_for(
                self.q.get(), timeout or self.client.timeout)
        except asyncio.TimeoutError as e:
            raise RpcTimeout(f"stream {self.code} recv timed out") from e
        if isinstance(reply, Exception):
            raise reply
        return reply.raise_if_error()

    async def call(self, header: dict | None = None, data: bytes = b"",
                   status: Status = Status.Running) -> Message:
        await self.send(header, data, status)
        return await self.recv()

    def close(self) -> None:
        self.client._pending.pop(self.req_id, None)


class ClientFactory:
    """Connection pool keyed by (host, port)."""

    def __init__(self, timeout_ms: int = 60_000, conns_per_addr: int = 1):
        self.timeout_ms = timeout_ms
        self.conns_per_addr = conns_per_addr
        self._pool: dict[tuple[str, int], list[RpcClient]] = {}
        self._rr: dict[tuple[str, int], int] = {}
        self._lock = asyncio.Lock()

    async def get(self, hostname: str, port: int) -> RpcClient:
        key = (hostname, port)
        async with self._lock:
            clients = self._pool.setdefault(key, [])
            clients[:] = [c for c in clients if c.is_connected]
            if len(clients) < self.conns_per_addr:
                c = await RpcClient(hostname, port, self.timeout_ms).connect()
                clients.append(c)
                return c
            i = self._rr.get(key, 0)
            self._rr[key] = i + 1
            return clients[i % len(clients)]

    async def invalidate(self, hostname: str, port: int) -> None:
        async with self._lock:
            for c in self._pool.pop((hostname, port), []):
                await c.close()

    async def close(self) -> None:
        async with self._lock:
            for clients in self._pool.values():
                for c in clients:
                    await c.close()
            self._pool.clear()


class ClusterConnector:
    """Master failover: tries each configured master address, remembers the
    leader, retries NotLeader / connection errors
    (client/cluster_connector.rs analog)."""

    def __init__(self, addrs: list[str], timeout_ms: int = 60_000, retries: int = 3):
        import uuid
        self.addrs = [(a.split(":")[0], int(a.split(":")[1])) for a in addrs]
        self.factory = ClientFactory(timeout_ms)
        self.retries = retries
        self._leader: Optional[tuple[str, int]] = None
        # client identity for the master's mutation retry-cache
        self.cid = uuid.uuid4().hex[:16]
        self._rid = 0

    async def rpc(self, code: RpcCode, header: dict | None = None,
                  data: bytes = b"", timeout: float | None = None) -> Message:
        # stable retry key: a re-sent logical call carries the same (cid,
        # rid) so the master's retry cache dedups replayed mutations
        header = dict(header or {})
        self._rid += 1
        header.setdefault("cid", self.cid)
        header.setdefault("rid", self._rid)
        last: Exception = ConnectError("no master addresses")
        order = ([self._leader] if self._leader else []) + \
                [a for a in self.addrs if a != self._leader]
        for attempt in range(self.retries):
            for addr in order:
                try:
                    client = await self.factory.get(*addr)
                    reply = await client.rpc(code, header, data, timeout)
                    self._leader = addr
                    return reply
                except NotLeader as e:
                    last = e
                    hint = str(e)
                    if ":" in hint:  # "leader=host:port" hint
                        for tok in hint.replace("=", " ").split():
                            if ":" in tok:
                                h, _, p = tok.rpartition(":")
                                if p.isdigit():
                                    self._leader = (h, int(p))
                                    order = [self._leader]
                                    break
                except (ConnectError, RpcTimeout) as e:
                    last = e
                    await self.factory.invalidate(*addr)
            self._leader = None
            order = list(self.addrs)
            if attempt + 1 < self.retries:
                await asyncio.sleep(min(0.1 * (2 ** attempt), 2.0))
        raise last

    async def close(self) -> None:
        await self.factory.close()
