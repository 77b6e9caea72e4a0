"""Wire format.

Frame layout follows the reference's 22-byte protocol block
(/root/reference/crates/core/rpc/src/message/rpc_message.rs:28-99):

    header_len: u32 BE | data_len: u32 BE | code: u8 | status: u8
    | req_id: u64 BE | seq_id: u32 BE | header bytes | data bytes

``status`` packs the streaming state: request state in the low nibble,
response state in the high nibble (rpc_message.rs:43-56).  Deviation from the
reference: headers are msgpack maps instead of protobuf messages (no protoc
in this environment); field names follow proto/master.proto & worker.proto.
"""
from __future__ import annotations

import itertools
import struct
from dataclasses import dataclass, field
from enum import IntEnum

import msgpack

from curvine_amd.errors import FsError
from curvine_amd.rpc.codes import RpcCode

PROTO_SIZE = 22
_HDR = struct.Struct(">IIBBQI")
MAX_DATA_SIZE = 16 << 20   # per-frame payload cap (rpc_message.rs:39)


class Status(IntEnum):
    """Streaming state machine (Open/Running/Complete/Cancel + error)."""
    Unary = 0
    Open = 1
    Running = 2
    Complete = 3
    Cancel = 4
    Error = 5

    @property
    def is_streaming(self) -> bool:
        return self in (Status.Open, Status.Running)


_STATUS_BY_VAL = [Status(i) if i in Status._value2member_map_ else Status.Unary
                  for i in range(16)]

_req_ids = itertools.count(1)


def next_req_id() -> int:
    return next(_req_ids)


@dataclass(slots=True)
class Message:
    code: int = 0
    req_status: Status = Status.Unary
    resp_status: Status = Status.Unary
    req_id: int = 0
    seq_id: int = 0
    header: dict = field(default_factory=dict)
    data: bytes = b""
    # undecodable-as-msgpack header bytes (candidate protobuf header from
    # a reference client; rpc/proto.py decides)
    raw_header: bytes = b""

    # ---------------- encode / decode ----------------
    def encode(self) -> bytes:
        hdr_bytes = self.raw_header or (
            msgpack.packb(self.header, use_bin_type=True) if self.header else b"")
        status = (int(self.resp_status) << 4) | int(self.req_status)
        proto = _HDR.pack(len(hdr_bytes), len(self.data), self.code & 0xFF,
                          status, self.req_id, self.seq_id)
        return proto + hdr_bytes + bytes(self.data)

    def encode_parts(self) -> list[bytes]:
        """Zero-copy-ish: [proto+header, data] so large payloads aren't copied."""
        hdr_bytes = self.raw_header or (
            msgpack.packb(self.header, use_bin_type=True) if self.header else b"")
        status = (int(self.resp_status) << 4) | int(self.req_status)
        proto = _HDR.pack(len(hdr_bytes), len(self.data), self.code & 0xFF,
                          status, self.req_id, self.seq_id)
        return [proto + hdr_bytes, self.data]

    @staticmethod
    def decode_proto(proto: bytes) -> tuple[int, int, "Message"]:
        hlen, dlen, code, status, req_id, seq_id = _HDR.unpack(proto)
        msg = Message(code=code,
                      req_status=_STATUS_BY_VAL[status & 0xF],
                      resp_status=_STATUS_BY_VAL[(status >> 4) & 0xF],
                      req_id=req_id, seq_id=seq_id)
        return hlen, dlen, msg

    def set_header_bytes(self, b: bytes) -> None:
        """Decode a header; non-msgpack bytes (a protobuf header from a
        reference peer, or garbage) are kept raw for the handler's
        per-connection codec to interpret."""
        if not b:
            self.header = {}
            return
        try:
            h = msgpack.unpackb(b, raw=False)
        except Exception:  # noqa: BLE001 — candidate protobuf header
            h = None
        if isinstance(h, dict):
            self.header = h
        else:
            self.header = {}
            self.raw_header = bytes(b)

    # ---------------- helpers ----------------
    @staticmethod
    def request(code: RpcCode, header: dict | None = None, data: bytes = b"",
                req_status: Status = Status.Unary, req_id: int | None = None,
                seq_id: int = 0) -> "Message":
        return Message(code=int(code), req_status=req_status,
                       req_id=next_req_id() if req_id is None else req_id,
                       seq_id=seq_id, header=header or {}, data=data)

    def reply(self, header: dict | None = None, data: bytes = b"",
              resp_status: Status = Status.Complete) -> "Message":
        return Message(code=self.code, req_status=self.req_status,
                       resp_status=resp_status, req_id=self.req_id,
                       seq_id=self.seq_id, header=header or {}, data=data)

    def error_reply(self, e: Exception) -> "Message":
        if isinstance(e, FsError):
            code, emsg = e.encode()
        else:
            code, emsg = 1, f"{type(e).__name__}: {e}"
        return Message(code=self.code, req_status=self.req_status,
                       resp_status=Status.Error, req_id=self.req_id,
                       seq_id=self.seq_id,
                       header={"error_code": code, "error_msg": emsg})

    @property
    def is_error(self) -> bool:
        return self.resp_status == Status.Error

    def raise_if_error(self) -> "Message":
        if self.is_error:
            raise FsError.decode(self.header.get("error_code", 1),
                                 self.header.get("error_msg", "rpc error"))
        return self
