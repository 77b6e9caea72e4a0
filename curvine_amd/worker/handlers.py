"""Worker RPC handlers: block read/write streams.

Analog of /root/reference/curvine-worker/src/worker/handler/
(block_handler.rs routing, read_handler.rs:42-243 streaming chunked reads,
write_handler.rs:28-310 open/write/commit, batch_write_handler.rs small
blocks).  Reads push chunks (TCP backpressure throttles); writes ack each
chunk so the client can run a pipelined window.
"""
from __future__ import annotations

import asyncio
import logging
import time
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.message import Message, Status

log = logging.getLogger("curvine.worker.handler")


class WorkerHandler:
    def __init__(self, worker):
        self.worker = worker
        self.store = worker.store
        # per-connection write sessions: req_id -> writer state
        self.writes: dict[int, dict] = {}

    async def on_close(self):
        # connection dropped mid-write: roll back reservations
        for sess in self.writes.values():
            try:
                self.store.abort(sess["block_id"])
            except Exception:  # noqa: BLE001
                pass
        self.writes.clear()

    async def handle(self, msg: Message, conn) -> Optional[Message]:
        code = msg.code
        if msg.raw_header:
            from curvine_amd.rpc import proto as _proto
            decoded = _proto.decode_request(code, msg.raw_header)
            if decoded is None:
                raise err.InvalidArgument(
                    f"undecodable header for code {code}")
            msg.header = decoded
            msg.raw_header = b""
            conn.state["pbuf"] = True
        if conn.state.get("pbuf"):
            # reference peers: errors ride the DATA section in the
            # ErrorEncoder layout; replies get protobuf headers
            from curvine_amd.rpc import proto as _proto
            from curvine_amd.rpc.message import Status
            try:
                out = await self._handle_inner(msg, conn)
            except Exception as e:  # noqa: BLE001
                out = msg.reply(resp_status=Status.Error)
                out.data = _proto.encode_error(e)
                return out
            if out is not None and out.header:
                enc = _proto.encode_response(code, out.header)
                if enc is not None:
                    out.header = {}
                    out.raw_header = enc
            return out
        return await self._handle_inner(msg, conn)

    async def _handle_inner(self, msg: Message, conn) -> Optional[Message]:
        code = msg.code
        if code == int(RpcCode.WriteBlock):
            return await self._write_block(msg)
        if code == int(RpcCode.ReadBlock):
            return await self._read_block(msg, conn)
        if code == int(RpcCode.WriteBlocksBatch):
            return await self._write_batch(msg)
        if code == int(RpcCode.ShortCircuitInfo):
            h = msg.header
            info = self.store.local_info(h["block_id"])
            if h.get("pin"):
                # hold a store reader so delete/demote defer while an
                # out-of-process short-circuit reader streams the extent;
                # leased — the heartbeat loop reaps expired pins
                import time as _t
                import uuid as _u
                r = self.store.open_reader(h["block_id"])
                token = _u.uuid4().hex
                lease_s = min(3600, int(h.get("lease_ms", 300_000)) / 1000)
                self.worker.pins[token] = (r, _t.monotonic() + lease_s)
                info["pin_token"] = token
            return msg.reply({"info": info})
        if code == int(RpcCode.UnpinBlock):
            ent = self.worker.pins.pop(msg.header.get("token", ""), None)
            if ent is not None:
                try:
                    ent[0].close()
                except Exception:  # noqa: BLE001
                    pass
            return msg.reply({})
        if code == int(RpcCode.Heartbeat):
            return msg.reply({"worker_id": self.worker.worker_id})
        raise err.Unsupported(f"worker rpc code {code}")

    # ---------------- write ----------------
    async def _write_block(self, msg: Message) -> Message:
        loop = asyncio.get_event_loop()
        if msg.req_status == Status.Open:
            h = msg.header
            if h.get("reopen"):
                # positional rewrite of an existing block (random writes)
                writer = await loop.run_in_executor(
                    None, self.store.reopen_writer, h["block_id"])
            else:
                writer = await loop.run_in_executor(
                    None, self.store.create_writer, h["block_id"],
                    h.get("reserve", 64 << 20), h.get("tier", ""))
            self.writes[msg.req_id] = {"writer": writer,
                                       "block_id": h["block_id"],
                                       "reopen": bool(h.get("reopen")),
                                       "t0": time.perf_counter()}
            return msg.reply({"ok": True}, resp_status=Status.Running)
        sess = self.writes.get(msg.req_id)
        if sess is None:
            raise err.FsError("write stream not open")
        if msg.req_status == Status.Running:
            if msg.data:
                off = msg.header.get("off")
                if off is not None:
                    await loop.run_in_executor(None, sess["writer"].pwrite,
                                               off, msg.data)
                else:
                    await loop.run_in_executor(None, sess["writer"].write,
                                               msg.data)
            return msg.reply({}, resp_status=Status.Running)
        if msg.req_status == Status.Complete:
            if msg.header.get("no_finalize") or (
                    sess["reopen"] and "length" not in msg.header):
                # in-place rewrite: block metadata (length/state) unchanged
                self.writes.pop(msg.req_id, None)
                return msg.reply({}, resp_status=Status.Complete)
            length = msg.header.get("length", sess["writer"].pos)
            tier = await loop.run_in_executor(
                None, self.store.finalize, sess["block_id"], length)
            self.writes.pop(msg.req_id, None)
            return msg.reply({"tier": tier,
                              "crc32c": self.store.block_crc(sess["block_id"])},
                             resp_status=Status.Complete)
        if msg.req_status == Status.Cancel:
            await loop.run_in_executor(None, self.store.abort, sess["block_id"])
            self.writes.pop(msg.req_id, None)
            return msg.reply({}, resp_status=Status.Complete)
        raise err.InvalidArgument(f"write stream status {msg.req_status}")

    async def _write_batch(self, msg: Message) -> Message:
        """Many small blocks in one frame: header {blocks: [{block_id,
        length, tier}...]}, data = concatenated payloads."""
        loop = asyncio.get_event_loop()
        data = msg.data
        off = 0
        done = []
        for b in msg.header.get("blocks", []):
            n = b["length"]
            writer = await loop.run_in_executor(
                None, self.store.create_writer, b["block_id"], n,
                b.get("tier", ""))
            await loop.run_in_executor(None, writer.write, data[off:off + n])
            await loop.run_in_executor(None, self.store.finalize,
                                       b["block_id"], n)
            off += n
            done.append(b["block_id"])
        return msg.reply({"committed": done})

    # ---------------- read ----------------
    async def _read_block(self, msg: Message, conn) -> Optional[Message]:
        if msg.req_status != Status.Open:
            return None   # pull-mode not used; server pushes
        h = msg.header
        loop = asyncio.get_event_loop()
        reader = await loop.run_in_executor(None, self.store.open_reader,
                                            h["block_id"])
        try:
            offset = h.get("offset", 0)
            length = min(h.get("length", reader.length),
                         reader.length - offset)
            chunk = h.get("chunk_size", 1 << 20)
            ack = msg.reply({"length": reader.length},
                            resp_status=Status.Running)
            if conn.state.get("pbuf"):
                # reference peers: BlockReadResponse protobuf open-ack
                from curvine_amd.rpc import proto as _proto
                enc = _proto.encode_response(msg.code, ack.header)
                if enc is not None:
                    ack.header = {}
                    ack.raw_header = enc
            await conn.send(ack)
            pos = 0
            t0 = time.perf_counter()
            # read wide spans per executor hop (the thread handoff costs
            # more than the copy), stream frame-sized slices zero-copy
            span_sz = max(chunk, 8 << 20)
            while pos < length:
                span = min(span_sz, length - pos)
                buf = bytearray(span)
                await loop.run_in_executor(None, reader.read_into,
                                           offset + pos, buf, 0, span)
                mv = memoryview(buf)
                sent = 0
                while sent < span:
                    n = min(chunk, span - sent)
                    reply = Message(code=msg.code, req_status=msg.req_status,
                                    resp_status=Status.Running,
                                    req_id=msg.req_id, seq_id=msg.seq_id,
                                    data=mv[sent:sent + n])
                    await conn.send(reply)
                    sent += n
                pos += span
            dt_us = (time.perf_counter() - t0) * 1e6
            if dt_us > self.worker.conf.worker.io_slow_us:
                log.warning("slow read block=%d len=%d %.0fus",
                            h["block_id"], length, dt_us)
            return msg.reply(resp_status=Status.Complete)
        finally:
            reader.close()
