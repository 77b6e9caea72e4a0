"""Per-block transfer clients.

Analog of /root/reference/crates/client/curvine-client-core/src/block/
(block_writer.rs remote writer adapters, block_reader_remote.rs streaming
reads, block_client_pool.rs pooled conns).  Three access paths:

* local in-process (worker registry) — the HBM short-circuit,
* remote streaming RPC (WriteBlock/ReadBlock codes),
* hole reader synthesizing zeros (block_reader_hole.rs analog).
"""
from __future__ import annotations

import asyncio
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.model import WorkerAddress
from curvine_amd.rpc.client import ClientFactory, RpcStream
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.message import Status

# one pool per event loop (tests spin up fresh loops)
_factories: dict[int, ClientFactory] = {}


def factory() -> ClientFactory:
    loop = asyncio.get_running_loop()
    f = _factories.get(id(loop))
    if f is None or getattr(f, "_loop_ref", None) is not loop:
        f = ClientFactory(conns_per_addr=2)
        f._loop_ref = loop  # type: ignore[attr-defined]
        _factories[id(loop)] = f
    return f


# ---------------------------------------------------------------------------
# Writers
# ---------------------------------------------------------------------------

class BlockWriterLocal:
    """Writes straight into the colocated worker's block store."""

    def __init__(self, store, block_id: int, reserve: int, tier: str,
                 reopen: bool = False):
        self.store = store
        self.reopen = reopen
        self.writer = (store.reopen_writer(block_id) if reopen
                       else store.create_writer(block_id, reserve, tier))
        self.block_id = block_id
        self.pos = 0

    async def write(self, data) -> None:
        loop = asyncio.get_event_loop()
        await loop.run_in_executor(None, self.writer.write, data)
        self.pos += len(data)

    async def pwrite(self, off: int, data) -> None:
        loop = asyncio.get_event_loop()
        await loop.run_in_executor(None, self.writer.pwrite, off, data)

    async def commit(self, length: int | None = None) -> str:
        if length is None:       # in-place rewrite: nothing to publish
            return ""
        loop = asyncio.get_event_loop()
        tier = await loop.run_in_executor(None, self.store.finalize,
                                          self.block_id, length)
        self.last_crc = self.store.block_crc(self.block_id)
        return tier

    async def abort(self) -> None:
        if not self.reopen:
            self.store.abort(self.block_id)


class BlockWriterRemote:
    """Streaming WriteBlock: bulk appends take the native C++ session
    (dw_open/dw_write in csrc/data_server.cpp — pooled conn, GIL-free
    sendmsg loop with a pipelined ack window, one executor hop per
    chunk); the asyncio stream remains for pwrite and as the fallback."""

    import os as _os
    WINDOW = int(_os.environ.get("CURVINE_DW_WINDOW", "4"))
    DW_CHUNK = int(_os.environ.get("CURVINE_DW_CHUNK", str(4 << 20)))
    del _os

    def __init__(self, addr: WorkerAddress, block_id: int, reserve: int,
                 tier: str, reopen: bool = False):
        self.addr = addr
        self.block_id = block_id
        self.reserve = reserve
        self.tier = tier
        self.reopen = reopen
        self.stream: Optional[RpcStream] = None
        self.inflight = 0
        self.pos = 0
        self._dw = None         # native session id
        self._dw_tried = False

    async def _ensure_open(self) -> None:
        if self.stream is not None:
            return
        client = await factory().get(self.addr.hostname, self.addr.rpc_port)
        self.stream = client.stream(RpcCode.WriteBlock)
        reply = await self.stream.call(
            {"block_id": self.block_id, "reserve": self.reserve,
             "tier": self.tier, "reopen": self.reopen}, status=Status.Open)

    async def _ensure_native(self) -> bool:
        """Open the native write session once; False -> asyncio path."""
        if self._dw is not None:
            return True
        if self._dw_tried or self.stream is not None:
            return False
        self._dw_tried = True
        lib = _native_data_lib()
        if lib is None or not hasattr(lib, "dw_open"):
            return False
        loop = asyncio.get_running_loop()
        hid, st, hdr = await loop.run_in_executor(
            None, lib.dw_open, self.addr.hostname, self.addr.rpc_port,
            self.block_id, self.reserve, self.tier, self.reopen,
            self.WINDOW)
        if st == 5 and hdr:
            _raise_wire_error(hdr)
        if hid <= 0:
            return False
        self._dw = hid
        return True

    async def write(self, data) -> None:
        if await self._ensure_native():
            lib = _native_data_lib()
            loop = asyncio.get_running_loop()
            ok = await loop.run_in_executor(
                None, lib.dw_write, self._dw, data, 0, len(data),
                self.DW_CHUNK)
            if not ok:
                lib.dw_abort(self._dw)
                self._dw = None
                raise ConnectError(
                    f"native write stream to {self.addr.hostname} failed")
            self.pos += len(data)
            return
        await self._ensure_open()
        # no bytes() copy: the transport copies what it cannot send
        # immediately, so a memoryview is safe to pass through
        await self.stream.send({}, data, Status.Running)
        self.inflight += 1
        self.pos += len(data)
        while self.inflight >= self.WINDOW:
            await self.stream.recv()
            self.inflight -= 1

    async def pwrite(self, off: int, data) -> None:
        if self._dw is not None:
            # keep the native append session; rewrites ride a separate
            # reopen stream.  Drain the append window first — the two
            # connections are unordered, so the bytes being rewritten
            # must already be consumed by the server
            lib = _native_data_lib()
            ok = await asyncio.get_running_loop().run_in_executor(
                None, lib.dw_drain, self._dw)
            if not ok:
                lib.dw_abort(self._dw)
                self._dw = None
                raise ConnectError(
                    f"native write stream to {self.addr.hostname} failed")
            if self.stream is None:
                client = await factory().get(self.addr.hostname,
                                             self.addr.rpc_port)
                self.stream = client.stream(RpcCode.WriteBlock)
                await self.stream.call(
                    {"block_id": self.block_id, "reserve": 0,
                     "tier": self.tier, "reopen": True},
                    status=Status.Open)
        await self._ensure_open()
        await self.stream.send({"off": off}, data, Status.Running)
        self.inflight += 1
        while self.inflight >= self.WINDOW:
            await self.stream.recv()
            self.inflight -= 1

    async def _dw_finish(self, length: int | None = None,
                         no_finalize: bool = False) -> dict:
        import msgpack
        lib = _native_data_lib()
        loop = asyncio.get_running_loop()
        hid, self._dw = self._dw, None
        st, hdr = await loop.run_in_executor(
            None, lib.dw_commit, hid, length if length is not None else -1,
            no_finalize)
        if st == 5:
            _raise_wire_error(hdr)
        try:
            return msgpack.unpackb(hdr, raw=False) if hdr else {}
        except Exception:  # noqa: BLE001
            return {}

    async def commit(self, length: int | None = None) -> str:
        if self._dw is not None:
            if self.stream is not None:
                # close the rewrite side stream first, without finalize
                while self.inflight > 0:
                    await self.stream.recv()
                    self.inflight -= 1
                await self.stream.send({"no_finalize": True}, b"",
                                       Status.Complete)
                await self.stream.recv()
                self.stream.close()
                self.stream = None
            h = await self._dw_finish(length, no_finalize=length is None)
            self.last_crc = h.get("crc32c")
            return h.get("tier", "")
        await self._ensure_open()
        while self.inflight > 0:
            await self.stream.recv()
            self.inflight -= 1
        hdr = {"no_finalize": True} if length is None else {"length": length}
        await self.stream.send(hdr, b"", Status.Complete)
        reply = await self.stream.recv()
        self.stream.close()
        self.last_crc = reply.header.get("crc32c")
        return reply.header.get("tier", "")

    def __del__(self):
        # a writer dropped without commit/abort must not leak its native
        # session (pooled fd + C++ entry); socket-only, safe from GC
        dw = getattr(self, "_dw", None)
        if dw is not None:
            try:
                lib = _native_data_lib()
                if lib is not None:
                    lib.dw_abort(dw)
            except Exception:  # noqa: BLE001 — interpreter teardown
                pass

    async def abort(self) -> None:
        if self._dw is not None:
            lib = _native_data_lib()
            hid, self._dw = self._dw, None
            try:
                await asyncio.get_running_loop().run_in_executor(
                    None, lib.dw_abort, hid)
            except Exception:  # noqa: BLE001
                pass
        if self.stream is not None:
            try:
                await self.stream.send({}, b"", Status.Cancel)
            except Exception:  # noqa: BLE001
                pass
            self.stream.close()


def make_block_writer(addr: WorkerAddress, block_id: int, reserve: int,
                      tier: str, reopen: bool = False):
    from curvine_amd.worker import registry
    store = registry.lookup(addr.worker_id)
    if store is not None:
        return BlockWriterLocal(store, block_id, reserve, tier, reopen)
    return BlockWriterRemote(addr, block_id, reserve, tier, reopen)


# ---------------------------------------------------------------------------
# Readers
# ---------------------------------------------------------------------------

class BlockReaderLocal:
    """In-process short-circuit read (HBM arena or file tier)."""

    def __init__(self, store, block_id: int):
        self.reader = store.open_reader(block_id)
        self.length = self.reader.length

    async def read_into(self, off: int, out, out_off: int, n: int) -> int:
        loop = asyncio.get_event_loop()
        return await loop.run_in_executor(
            None, self.reader.read_into, off, out, out_off, n)

    async def read(self, off: int, n: int) -> bytes:
        loop = asyncio.get_event_loop()
        return await loop.run_in_executor(None, self.reader.read, off, n)

    def read_into_sync(self, off: int, out, out_off: int, n: int) -> int:
        return self.reader.read_into(off, out, out_off, n)

    async def read_to_device(self, off: int, dst_ptr: int, n: int) -> int:
        loop = asyncio.get_event_loop()
        return await loop.run_in_executor(
            None, self.reader.read_to_ptr, off, dst_ptr, n, True)

    def close(self) -> None:
        self.reader.close()


def _native_data_lib():
    """The C++ data-plane client (csrc/data_server.cpp data_read_into /
    data_write_from), or None when the extension lacks it."""
    global _NATIVE_DATA
    if _NATIVE_DATA is not False:
        return _NATIVE_DATA
    try:
        from curvine_amd import native
        lib = native.load()
        _NATIVE_DATA = lib if hasattr(lib, "data_read_into") else None
    except Exception:  # noqa: BLE001 — extension unavailable
        _NATIVE_DATA = None
    return _NATIVE_DATA


_NATIVE_DATA: object = False


def _raise_wire_error(hdr_bytes: bytes) -> None:
    import msgpack

    from curvine_amd.errors import FsError
    try:
        h = msgpack.unpackb(hdr_bytes, raw=False) if hdr_bytes else {}
    except Exception:  # noqa: BLE001
        h = {}
    raise FsError.decode(h.get("error_code", 1),
                         h.get("error_msg", "data stream error"))


class BlockReaderRemote:
    """Streaming ReadBlock: the server pushes chunks; we buffer in-order.

    Bulk reads take the native C++ client path (blocking socket loop with
    the GIL released, chunks recv'd straight into the destination buffer);
    the asyncio iterator remains for chunk-streaming consumers and as the
    fallback."""

    def __init__(self, addr: WorkerAddress, block_id: int):
        self.addr = addr
        self.block_id = block_id
        self.length = 0

    async def read_range(self, off: int, n: int, chunk_size: int = 4 << 20):
        """Async iterator of chunks covering [off, off+n)."""
        client = await factory().get(self.addr.hostname, self.addr.rpc_port)
        stream = client.stream(RpcCode.ReadBlock)
        reply = await stream.call({"block_id": self.block_id, "offset": off,
                                   "length": n, "chunk_size": chunk_size},
                                  status=Status.Open)
        self.length = reply.header.get("length", 0)
        try:
            while True:
                m = await stream.recv()
                if m.resp_status == Status.Complete:
                    break
                yield m.data
        finally:
            stream.close()

    async def read_into(self, off: int, out, out_off: int, n: int) -> int:
        lib = _native_data_lib()
        if lib is not None:
            try:
                loop = asyncio.get_event_loop()
                status, hdr, got = await loop.run_in_executor(
                    None, lib.data_read_into, self.addr.hostname,
                    self.addr.rpc_port, self.block_id, off, n, out, out_off,
                    4 << 20)
                if status == int(Status.Error):
                    _raise_wire_error(hdr)
                if status == int(Status.Complete):
                    return got
                # torn stream: fall through to the asyncio path
            except (RuntimeError, TypeError):
                pass   # connect failed / non-buffer dst: asyncio fallback
        got = 0
        async for chunk in self.read_range(off, n):
            out[out_off + got:out_off + got + len(chunk)] = chunk
            got += len(chunk)
        return got

    async def read(self, off: int, n: int) -> bytes:
        lib = _native_data_lib()
        if lib is not None:
            buf = bytearray(n)
            got = await self.read_into(off, buf, 0, n)
            return bytes(memoryview(buf)[:got])
        parts = []
        async for chunk in self.read_range(off, n):
            parts.append(chunk)
        return b"".join(parts)

    def close(self) -> None:
        pass


class BlockReaderHole:
    """Sparse hole: synthesizes zeros (block_reader_hole.rs analog)."""

    def __init__(self, length: int):
        self.length = length

    async def read_into(self, off: int, out, out_off: int, n: int) -> int:
        n = max(0, min(n, self.length - off))
        out[out_off:out_off + n] = b"\x00" * n
        return n

    async def read(self, off: int, n: int) -> bytes:
        return b"\x00" * max(0, min(n, self.length - off))

    def close(self) -> None:
        pass


# --------------------------------------------------------------------------
# Cross-process HBM short-circuit (hipIpc)
# --------------------------------------------------------------------------
#
# A colocated worker in ANOTHER process discloses its device arena via
# ShortCircuitInfo (hipIpc handle + extent); the client maps the arena
# once per (worker, arena) and serves reads with direct D2H DMA in its
# own process — no socket in the byte path.  The worker pins the block
# (store reader lease) while this reader lives, so delete/demote defer
# exactly like the in-process short circuit.

class _IpcArenaPool:
    """(host, port, arena_handle) -> mapped native Arena, per process."""

    def __init__(self):
        import threading
        self._mu = threading.Lock()
        self._arenas: dict = {}

    def get(self, key, ipc_handle: bytes, cap: int, device: int):
        from curvine_amd import native
        with self._mu:
            a = self._arenas.get(key)
            if a is None:
                a = native.Arena.from_ipc(ipc_handle, cap, device)
                self._arenas[key] = a
            return a


_IPC_POOL = _IpcArenaPool()

_LOCAL_HOSTS = None


def _is_local_host(hostname: str) -> bool:
    global _LOCAL_HOSTS
    if _LOCAL_HOSTS is None:
        import socket
        _LOCAL_HOSTS = {"127.0.0.1", "localhost", socket.gethostname()}
    return hostname in _LOCAL_HOSTS


class BlockReaderIpc:
    """Store-reader-compatible view of a block living in ANOTHER
    process's HBM arena (same interface as BlockStore.open_reader's
    result: read_into/read_to_ptr/crc32c/meta/layout.arena/close)."""

    def __init__(self, arena, info: dict, unpin):
        from types import SimpleNamespace
        self._arena = arena
        self._off = info["offset"]
        self.length = info["length"]
        self.meta = {"kind": "arena", "offset": info["offset"],
                     "length": info["length"]}
        self.layout = SimpleNamespace(arena=arena)
        self._unpin = unpin

    def read_into(self, off: int, out, out_off: int, n: int) -> int:
        n = min(n, self.length - off)
        if n <= 0:
            return 0
        self._arena.read(self._off + off, out, out_off, n)
        return n

    def read_to_ptr(self, off: int, ptr: int, n: int, device: bool) -> int:
        n = min(n, self.length - off)
        if n <= 0:
            return 0
        self._arena.read_to_ptr(self._off + off, ptr, n, device)
        return n

    def crc32c(self, off: int, n: int) -> int:
        return self._arena.crc32c(self._off + off, n)

    async def read(self, off: int, n: int) -> bytes:
        out = bytearray(min(n, self.length - off))
        self.read_into(off, out, 0, len(out))
        return bytes(out)

    async def read_into_async(self, off: int, out, out_off: int, n: int):
        return self.read_into(off, out, out_off, n)

    def close(self) -> None:
        u, self._unpin = self._unpin, None
        if u is not None:
            try:
                u()
            except Exception:  # noqa: BLE001
                pass


class AsyncIpcReader:
    """Async adapter over BlockReaderIpc for the FsReader block-reader
    interface (DMA runs on an executor thread)."""

    def __init__(self, r: "BlockReaderIpc"):
        self._r = r
        self.length = r.length

    async def read_into(self, off: int, out, out_off: int, n: int) -> int:
        loop = asyncio.get_running_loop()
        return await loop.run_in_executor(None, self._r.read_into,
                                          off, out, out_off, n)

    def close(self) -> None:
        self._r.close()


async def open_ipc_reader(client, addr: WorkerAddress,
                          block_id: int):
    """Map a colocated (same-host, other-process) worker's HBM block via
    hipIpc, pinning it on the worker.  None when not applicable."""
    from curvine_amd import native
    if not _is_local_host(addr.hostname) or not native.gpu_available():
        return None
    try:
        c = await factory().get(addr.hostname, addr.rpc_port)
        r = await c.rpc(RpcCode.ShortCircuitInfo,
                        {"block_id": block_id, "pin": True})
        info = r.header.get("info", {})
        if info.get("kind") != "arena" or not info.get("ipc"):
            tok = info.get("pin_token")
            if tok:
                await c.rpc(RpcCode.UnpinBlock, {"token": tok})
            return None
        arena = _IPC_POOL.get((addr.hostname, addr.rpc_port,
                               info["arena_handle"]),
                              info["ipc"], info["cap"], info["device"])
        token = info.get("pin_token")

        def unpin():
            async def go():
                try:
                    cc = await factory().get(addr.hostname, addr.rpc_port)
                    await cc.rpc(RpcCode.UnpinBlock, {"token": token})
                except Exception:  # noqa: BLE001
                    pass
            try:
                asyncio.get_running_loop().create_task(go())
            except RuntimeError:
                pass   # no loop on this thread: the lease expires
        return BlockReaderIpc(arena, info, unpin if token else None)
    except Exception as e:  # noqa: BLE001 — fall back to the remote stream
        import logging
        logging.getLogger("curvine.client").debug(
            "ipc short-circuit unavailable for block %d at %s: %s",
            block_id, addr.hostname, e)
        return None
