"""Protobuf wire compatibility with the reference's RPC headers.

The reference frames are: 22-byte protocol block | protobuf header | raw
data, with header messages defined in
/root/reference/crates/common/curvine-proto/proto/{common,master,worker}
.proto (prost-generated, rpc_message.rs:28-99).  This module rebuilds the
SAME message schema (identical field names/numbers/types, transcribed
from those .proto files) at import time using google.protobuf's
descriptor machinery — there is no protoc in the image — and converts
between those messages and the internal msgpack header dicts.

Negotiation is per-connection and automatic: our peers send msgpack
headers (the fast native path); a reference client's first frame fails
msgpack decoding and parses as the code's protobuf request type, which
flips the connection to protobuf replies.  msgpack peers are untouched.

Deviations (documented, lossless where it matters):
* the HBM storage tier has no reference enum value and is reported as
  STORAGE_TYPE_PROTO_MEM on the wire;
* FileStatusProto.owner/group carry str(uid)/str(gid);
* ExtendedBlockProto.block_size carries the block's byte length (the
  reference derives per-block lengths from offsets; we do the inverse on
  decode).
"""
from __future__ import annotations

from typing import Callable, Optional

from google.protobuf import descriptor_pb2, descriptor_pool, message_factory

from curvine_amd.rpc.codes import RpcCode

F = descriptor_pb2.FieldDescriptorProto
_T = {
    "int32": F.TYPE_INT32, "int64": F.TYPE_INT64, "uint32": F.TYPE_UINT32,
    "uint64": F.TYPE_UINT64, "bool": F.TYPE_BOOL, "string": F.TYPE_STRING,
    "bytes": F.TYPE_BYTES, "enum": F.TYPE_ENUM, "msg": F.TYPE_MESSAGE,
}
_L = {"optional": F.LABEL_OPTIONAL, "required": F.LABEL_REQUIRED,
      "repeated": F.LABEL_REPEATED}

# (name, number, type, label[, type_name]) — type_name for enum/msg refs
_ENUMS = {
    "StorageTypeProto": ["MEM", "SSD", "HDD", "UFS", "DISK", "SPDK_DISK"],
    "TtlActionProto": ["NONE", "DELETE", "FREE"],
    "FileTypeProto": ["DIR", "FILE", "LINK", "STREAM", "AGG", "OBJECT",
                      "FIFO", "CHAR", "BLOCK", "SOCKET"],
    "WriteTypeProto": ["CACHE_MODE", "FS_MODE"],
}
# proto2 enums whose first value is nonzero
_ENUMS_BASED1 = {
    "StorageStateProto": ["CV", "UFS", "BOTH"],
    "BlockReportStatusProto": ["FINALIZED", "WRITING", "DELETED"],
}
_ENUMS0 = {"HeartbeatStatusProto": ["START", "RUNNING", "END"],
           # UNKNOWN is the zero value; consumers treat it as DIAGNOSE
           "CompatibilityModeProto": ["UNKNOWN", "DIAGNOSE", "ENFORCE"]}

_MESSAGES = {
    # ---- common.proto
    "StoragePolicyProto": [
        ("storage_type", 1, "enum", "required", "StorageTypeProto"),
        ("ttl_ms", 2, "int64", "required"),
        ("ttl_action", 3, "enum", "required", "TtlActionProto"),
        ("ufs_mtime", 4, "int64", "required"),
        ("state", 5, "enum", "required", "StorageStateProto"),
    ],
    "FileStatusProto": [
        ("id", 1, "int64", "required"),
        ("path", 2, "string", "required"),
        ("name", 3, "string", "required"),
        ("is_dir", 4, "bool", "required"),
        ("mtime", 5, "int64", "required"),
        ("atime", 6, "int64", "required"),
        ("children_num", 7, "int32", "required"),
        ("is_complete", 8, "bool", "required"),
        ("len", 9, "int64", "required"),
        ("replicas", 10, "int32", "required"),
        ("block_size", 11, "int64", "required"),
        ("file_type", 12, "enum", "required", "FileTypeProto"),
        ("x_attr", 13, "map_string_bytes", "repeated"),
        ("storage_policy", 14, "msg", "required", "StoragePolicyProto"),
        ("owner", 15, "string", "required"),
        ("group", 16, "string", "required"),
        ("mode", 17, "uint32", "required"),
        ("target", 18, "string", "optional"),
        ("nlink", 19, "uint32", "required"),
        ("ctime", 20, "int64", "optional"),
    ],
    "WorkerAddressProto": [
        ("worker_id", 1, "uint32", "required"),
        ("hostname", 2, "string", "required"),
        ("ip_addr", 3, "string", "required"),
        ("rpc_port", 4, "uint32", "required"),
        ("web_port", 5, "uint32", "required"),
    ],
    "StorageInfoProto": [
        ("dir_id", 1, "uint32", "required"),
        ("storage_id", 2, "string", "required"),
        ("failed", 3, "bool", "required"),
        ("capacity", 4, "int64", "required"),
        ("available", 5, "int64", "required"),
        ("fs_used", 6, "int64", "required"),
        ("non_fs_used", 7, "int64", "required"),
        ("reserved_bytes", 8, "int64", "required"),
        ("storage_type", 9, "enum", "required", "StorageTypeProto"),
        ("block_num", 10, "int64", "required"),
        ("dir_path", 11, "string", "required"),
    ],
    "ComponentInfoProto": [
        ("component", 1, "string", "optional"),
        ("release_version", 2, "string", "optional"),
        ("git_commit", 3, "string", "optional"),
        ("git_tag", 4, "string", "optional"),
        ("git_branch", 5, "string", "optional"),
        ("protocol_version", 6, "uint32", "optional"),
        ("min_protocol_version", 7, "uint32", "optional"),
        ("capabilities", 8, "string", "repeated"),
    ],
    "ServerCompatibilityInfoProto": [
        ("server", 1, "msg", "required", "ComponentInfoProto"),
        ("min_worker_version", 2, "string", "optional"),
        ("min_client_version", 3, "string", "optional"),
        ("compatibility_mode", 4, "enum", "required",
         "CompatibilityModeProto"),
        ("blocked_versions", 5, "string", "repeated"),
    ],
    "WorkerInfoProto": [
        ("address", 1, "msg", "required", "WorkerAddressProto"),
        ("capacity", 2, "int64", "required"),
        ("available", 3, "int64", "required"),
        ("fs_used", 4, "int64", "required"),
        ("non_fs_used", 5, "int64", "required"),
        ("last_update", 6, "uint64", "required"),
        ("storage_map", 7, "map_string_msg", "repeated",
         "StorageInfoProto"),
        ("reserved_bytes", 8, "int64", "required"),
        ("component_info", 1000, "msg", "optional", "ComponentInfoProto"),
    ],
    "GetFilesystemInfoRequest": [
        ("component_info", 1000, "msg", "optional", "ComponentInfoProto"),
    ],
    "GetFilesystemInfoResponse": [
        ("active_master", 1, "string", "required"),
        ("journal_nodes", 2, "string", "repeated"),
        ("inode_dir_num", 3, "int64", "required"),
        ("inode_file_num", 4, "int64", "required"),
        ("block_num", 5, "int64", "required"),
        ("capacity", 6, "int64", "required"),
        ("available", 7, "int64", "required"),
        ("fs_used", 8, "int64", "required"),
        ("non_fs_used", 9, "int64", "required"),
        ("reserved_bytes", 10, "int64", "required"),
        ("live_workers", 11, "msg", "repeated", "WorkerInfoProto"),
        ("blacklist_workers", 12, "msg", "repeated", "WorkerInfoProto"),
        ("decommission_workers", 13, "msg", "repeated", "WorkerInfoProto"),
        ("lost_workers", 14, "msg", "repeated", "WorkerInfoProto"),
        ("allocatable_capacity", 15, "int64", "optional"),
        ("allocatable_available", 16, "int64", "optional"),
        ("compatibility", 1000, "msg", "optional",
         "ServerCompatibilityInfoProto"),
    ],
    "FileAllocOptsProto": [
        ("truncate", 1, "bool", "required"),
        ("off", 2, "int64", "required"),
        ("len", 3, "int64", "required"),
        ("mode", 4, "int32", "required"),
    ],
    "ExtendedBlockProto": [
        ("id", 1, "int64", "required"),
        ("block_size", 2, "int64", "required"),
        ("storage_type", 3, "enum", "required", "StorageTypeProto"),
        ("file_type", 4, "enum", "required", "FileTypeProto"),
        ("alloc_opts", 5, "msg", "optional", "FileAllocOptsProto"),
    ],
    "LocatedBlockProto": [
        ("block", 1, "msg", "required", "ExtendedBlockProto"),
        ("offset", 2, "int64", "required"),
        ("locs", 3, "msg", "repeated", "WorkerAddressProto"),
        ("has_spdk", 4, "bool", "optional"),
    ],
    "FileBlocksProto": [
        ("status", 1, "msg", "required", "FileStatusProto"),
        ("block_locs", 2, "msg", "repeated", "LocatedBlockProto"),
    ],
    "ClientAddressProto": [
        ("client_name", 1, "string", "required"),
        ("hostname", 2, "string", "required"),
        ("ip_addr", 3, "string", "required"),
        ("port", 4, "int32", "required"),
    ],
    "BlockLocationProto": [
        ("worker_id", 1, "uint32", "required"),
        ("storage_type", 2, "enum", "required", "StorageTypeProto"),
    ],
    "CommitBlockProto": [
        ("block_id", 1, "int64", "required"),
        ("block_len", 2, "int64", "required"),
        ("locations", 3, "msg", "repeated", "BlockLocationProto"),
    ],
    "FreeResultProto": [
        ("inodes", 2, "int64", "required"),
        ("bytes", 3, "int64", "required"),
    ],
    # ---- master.proto
    "MkdirOptsProto": [
        ("create_parent", 1, "bool", "required"),
        ("mode", 2, "uint32", "required"),
        ("x_attr", 3, "map_string_bytes", "repeated"),
        ("storage_policy", 4, "msg", "required", "StoragePolicyProto"),
        ("owner", 5, "string", "required"),
        ("group", 6, "string", "required"),
    ],
    "MkdirRequest": [
        ("path", 1, "string", "required"),
        ("opts", 2, "msg", "required", "MkdirOptsProto"),
    ],
    "MkdirResponse": [
        ("flag", 1, "bool", "required"),
        ("status", 2, "msg", "required", "FileStatusProto"),
    ],
    "CreateFileOptsProto": [
        ("create_flag", 1, "int32", "required"),
        ("create_parent", 2, "bool", "required"),
        ("file_type", 3, "enum", "required", "FileTypeProto"),
        ("replicas", 4, "int32", "required"),
        ("block_size", 5, "int64", "required"),
        ("x_attr", 6, "map_string_bytes", "repeated"),
        ("storage_policy", 7, "msg", "required", "StoragePolicyProto"),
        ("client_name", 8, "string", "required"),
        ("mode", 9, "uint32", "required"),
        ("owner", 10, "string", "required"),
        ("group", 11, "string", "required"),
        ("sync_ufs_meta", 12, "bool", "required"),
        ("ufs_len", 13, "int64", "required"),
    ],
    "CreateFileRequest": [
        ("path", 1, "string", "required"),
        ("opts", 2, "msg", "required", "CreateFileOptsProto"),
        ("flags", 3, "uint32", "required"),
    ],
    "CreateFileResponse": [
        ("file_status", 1, "msg", "required", "FileStatusProto"),
    ],
    "OpenFileRequest": [
        ("path", 1, "string", "required"),
        ("opts", 2, "msg", "required", "CreateFileOptsProto"),
        ("flags", 3, "uint32", "required"),
    ],
    "OpenFileResponse": [
        ("file_blocks", 1, "msg", "required", "FileBlocksProto"),
    ],
    "DeleteRequest": [
        ("path", 1, "string", "required"),
        ("recursive", 2, "bool", "required"),
    ],
    "DeleteResponse": [
        ("res", 1, "msg", "optional", "FreeResultProto"),
    ],
    "GetFileStatusRequest": [("path", 1, "string", "required")],
    "GetFileStatusResponse": [
        ("status", 1, "msg", "required", "FileStatusProto"),
    ],
    "ExistsRequest": [("path", 1, "string", "required")],
    "ExistsResponse": [("exists", 1, "bool", "required")],
    "ListStatusRequest": [
        ("path", 1, "string", "required"),
        ("need_location", 2, "bool", "required"),
    ],
    "ListStatusResponse": [
        ("statuses", 1, "msg", "repeated", "FileStatusProto"),
    ],
    "RenameRequest": [
        ("src", 1, "string", "required"),
        ("dst", 2, "string", "required"),
        ("flags", 3, "uint32", "required"),
    ],
    "RenameResponse": [("result", 1, "bool", "required")],
    "AddBlockRequest": [
        ("path", 1, "string", "required"),
        ("commit_blocks", 2, "msg", "repeated", "CommitBlockProto"),
        ("exclude_workers", 3, "uint32", "repeated"),
        ("located", 4, "bool", "required"),
        ("client_address", 5, "msg", "required", "ClientAddressProto"),
        ("file_len", 6, "int64", "required"),
        ("last_block", 7, "msg", "optional", "ExtendedBlockProto"),
        ("inode_id", 8, "int64", "optional"),
    ],
    "AddBlockResponse": [
        ("block", 1, "msg", "required", "LocatedBlockProto"),
    ],
    "CompleteFileRequest": [
        ("path", 1, "string", "required"),
        ("len", 2, "int64", "required"),
        ("client_name", 3, "string", "required"),
        ("commit_blocks", 4, "msg", "repeated", "CommitBlockProto"),
        ("only_flush", 5, "bool", "required"),
        ("inode_id", 6, "int64", "optional"),
        ("return_file_blocks", 8, "bool", "optional"),
    ],
    "CompleteFileResponse": [
        ("result", 1, "bool", "required"),
        ("file_blocks", 2, "msg", "optional", "FileBlocksProto"),
    ],
    "GetBlockLocationsRequest": [("path", 1, "string", "required")],
    "GetBlockLocationsResponse": [
        ("blocks", 1, "msg", "required", "FileBlocksProto"),
    ],
    "FreeRequest": [
        ("path", 1, "string", "required"),
        ("recursive", 2, "bool", "required"),
    ],
    "FreeResponse": [
        ("res", 1, "msg", "required", "FreeResultProto"),
    ],
    # ---- worker.proto
    "BlockWriteRequest": [
        ("block", 1, "msg", "required", "ExtendedBlockProto"),
        ("off", 2, "int64", "required"),
        ("block_size", 3, "int64", "required"),
        ("short_circuit", 4, "bool", "required"),
        ("client_name", 5, "string", "required"),
        ("chunk_size", 6, "int32", "required"),
        ("pipeline_stream", 7, "msg", "repeated", "WorkerAddressProto"),
    ],
    "BlockWriteResponse": [
        ("id", 1, "int64", "required"),
        ("path", 2, "string", "optional"),
        ("off", 3, "int64", "required"),
        ("block_size", 4, "int64", "required"),
        ("storage_type", 5, "enum", "required", "StorageTypeProto"),
    ],
    "BlockReadRequest": [
        ("id", 1, "int64", "required"),
        ("off", 2, "int64", "required"),
        ("len", 3, "int64", "required"),
        ("chunk_size", 4, "int32", "required"),
        ("short_circuit", 5, "bool", "required"),
        ("enable_read_ahead", 8, "bool", "required"),
        ("read_ahead_len", 9, "int64", "required"),
        ("drop_cache_len", 10, "int64", "required"),
    ],
    "BlockReadResponse": [
        ("id", 1, "int64", "required"),
        ("len", 2, "int64", "required"),
        ("path", 3, "string", "optional"),
        ("storage_type", 4, "enum", "required", "StorageTypeProto"),
    ],
}


def _build():
    fdp = descriptor_pb2.FileDescriptorProto()
    fdp.name = "curvine_wire.proto"
    fdp.package = "proto"
    fdp.syntax = "proto2"
    for name, vals in {**_ENUMS, **_ENUMS0}.items():
        e = fdp.enum_type.add()
        e.name = name
        prefix = _enum_prefix(name)
        for i, v in enumerate(vals):
            ev = e.value.add()
            ev.name = f"{prefix}_{v}"
            ev.number = i
    for name, vals in _ENUMS_BASED1.items():
        e = fdp.enum_type.add()
        e.name = name
        prefix = _enum_prefix(name)
        for i, v in enumerate(vals):
            ev = e.value.add()
            ev.name = f"{prefix}_{v}"
            ev.number = i + 1
    for mname, fields in _MESSAGES.items():
        m = fdp.message_type.add()
        m.name = mname
        for spec in fields:
            fname, num, ftype, label = spec[0], spec[1], spec[2], spec[3]
            fld = m.field.add()
            fld.name = fname
            fld.number = num
            fld.label = _L[label]
            if ftype == "map_string_msg":
                # map<string, Msg>: wire-identical to a repeated nested
                # entry message with the map_entry option
                entry = m.nested_type.add()
                entry.name = _camel(fname) + "Entry"
                entry.options.map_entry = True
                k = entry.field.add()
                k.name, k.number = "key", 1
                k.label, k.type = F.LABEL_OPTIONAL, F.TYPE_STRING
                v = entry.field.add()
                v.name, v.number = "value", 2
                v.label, v.type = F.LABEL_OPTIONAL, F.TYPE_MESSAGE
                v.type_name = f".proto.{spec[4]}"
                fld.label = F.LABEL_REPEATED
                fld.type = F.TYPE_MESSAGE
                fld.type_name = f".proto.{mname}.{entry.name}"
            elif ftype == "map_string_bytes":
                # map<string, bytes>: nested auto-generated entry message
                entry = m.nested_type.add()
                entry.name = _camel(fname) + "Entry"
                entry.options.map_entry = True
                k = entry.field.add()
                k.name, k.number = "key", 1
                k.label, k.type = F.LABEL_OPTIONAL, F.TYPE_STRING
                v = entry.field.add()
                v.name, v.number = "value", 2
                v.label, v.type = F.LABEL_OPTIONAL, F.TYPE_BYTES
                fld.label = F.LABEL_REPEATED
                fld.type = F.TYPE_MESSAGE
                fld.type_name = f".proto.{mname}.{entry.name}"
            else:
                fld.type = _T[ftype]
                if ftype in ("enum", "msg"):
                    fld.type_name = f".proto.{spec[4]}"
    pool = descriptor_pool.DescriptorPool()
    fd = pool.Add(fdp)
    out = {}
    for mname in _MESSAGES:
        out[mname] = message_factory.GetMessageClass(
            fd.message_types_by_name[mname])
    return out


def _enum_prefix(name: str) -> str:
    # FooBarProto -> FOO_BAR_PROTO (prost/protoc enum value prefixing)
    out = []
    for i, ch in enumerate(name):
        if ch.isupper() and i > 0:
            out.append("_")
        out.append(ch.upper())
    return "".join(out)


def _camel(snake: str) -> str:
    return "".join(p.capitalize() for p in snake.split("_"))


M = _build()

# ---------------------------------------------------------------- tiers

_TIER_TO_WIRE = {"MEM": 0, "SSD": 1, "HDD": 2, "UFS": 3, "DISK": 4,
                 "HBM": 0}  # HBM has no reference value: closest is MEM
_WIRE_TO_TIER = {0: "MEM", 1: "SSD", 2: "HDD", 3: "UFS", 4: "SSD", 5: "SSD"}
_FT_TO_WIRE = {0: 1, 1: 0, 2: 2}   # ours FILE=0,DIR=1,SYMLINK=2 -> proto
_FT_FROM_WIRE = {1: 0, 0: 1, 2: 2}

_O_CREAT, _O_TRUNC = 0x40, 0x200


def _policy(msg, tier: str, ttl_ms: int = 0, ttl_action: str = "none"):
    msg.storage_type = _TIER_TO_WIRE.get(tier, 0)
    msg.ttl_ms = ttl_ms
    msg.ttl_action = {"none": 0, "delete": 1, "free": 2}.get(ttl_action, 0)
    msg.ufs_mtime = 0
    msg.state = 1


def _status_to_proto(st: dict, msg) -> None:
    msg.id = st.get("inode_id", 0)
    msg.path = st.get("path", "/")
    msg.name = st.get("name", "")
    ft = st.get("file_type", 0)
    msg.is_dir = ft == 1
    msg.mtime = st.get("mtime_ms", 0)
    msg.atime = st.get("atime_ms", 0)
    msg.children_num = 0
    msg.is_complete = bool(st.get("is_complete", True))
    msg.len = st.get("length", 0)
    msg.replicas = st.get("replicas", 1)
    msg.block_size = st.get("block_size", 0)
    msg.file_type = _FT_TO_WIRE.get(ft, 1)
    for k, v in (st.get("xattrs") or {}).items():
        msg.x_attr[k] = bytes(v)
    _policy(msg.storage_policy, st.get("storage_tier", "MEM"),
            st.get("ttl_ms", 0), st.get("ttl_action", "none"))
    msg.owner = str(st.get("uid", 0))
    msg.group = str(st.get("gid", 0))
    msg.mode = st.get("mode", 0o644)
    if st.get("symlink_target"):
        msg.target = st["symlink_target"]
    msg.nlink = st.get("nlink", 1)


def _status_from_proto(msg) -> dict:
    def _int(s, default=0):
        try:
            return int(s)
        except (TypeError, ValueError):
            return default
    return {
        "inode_id": msg.id, "path": msg.path, "name": msg.name,
        "file_type": _FT_FROM_WIRE.get(msg.file_type, 0),
        "length": msg.len, "is_complete": msg.is_complete,
        "block_size": msg.block_size, "replicas": msg.replicas,
        "storage_tier": _WIRE_TO_TIER.get(
            msg.storage_policy.storage_type, "MEM"),
        "mtime_ms": msg.mtime, "atime_ms": msg.atime, "mode": msg.mode,
        "uid": _int(msg.owner), "gid": _int(msg.group),
        "ttl_ms": msg.storage_policy.ttl_ms,
        "ttl_action": {0: "none", 1: "delete", 2: "free"}.get(
            msg.storage_policy.ttl_action, "none"),
        "symlink_target": msg.target if msg.HasField("target") else "",
        "nlink": msg.nlink,
        "xattrs": {k: bytes(v) for k, v in msg.x_attr.items()},
    }


def _addr_to_proto(a: dict, msg) -> None:
    msg.worker_id = a.get("worker_id", 0) & 0xFFFFFFFF
    msg.hostname = a.get("hostname", "")
    msg.ip_addr = a.get("hostname", "")
    msg.rpc_port = a.get("rpc_port", 0)
    msg.web_port = a.get("web_port", 0) or 0


def _addr_from_proto(msg) -> dict:
    return {"worker_id": msg.worker_id, "hostname": msg.hostname,
            "rpc_port": msg.rpc_port, "web_port": msg.web_port}


def _file_blocks_to_proto(fb: dict, msg) -> None:
    _status_to_proto(fb.get("status", {}), msg.status)
    st = fb.get("status", {})
    for b in fb.get("blocks", []):
        lb = msg.block_locs.add()
        lb.block.id = b["block"]["block_id"]
        lb.block.block_size = b["block"].get("length", 0)
        tiers = b.get("tiers") or ["MEM"]
        lb.block.storage_type = _TIER_TO_WIRE.get(tiers[0], 0)
        lb.block.file_type = 1
        lb.offset = b.get("offset", 0)
        for a in b.get("locations", []):
            _addr_to_proto(a, lb.locs.add())
    _ = st


def _file_blocks_from_proto(msg) -> dict:
    blocks = []
    for lb in msg.block_locs:
        tier = _WIRE_TO_TIER.get(lb.block.storage_type, "MEM")
        blocks.append({
            "block": {"block_id": lb.block.id,
                      "length": lb.block.block_size, "state": 1},
            "offset": lb.offset,
            "locations": [_addr_from_proto(a) for a in lb.locs],
            "tiers": [tier] * len(lb.locs),
        })
    return {"status": _status_from_proto(msg.status), "blocks": blocks}


# ------------------------------------------------------------ converters
# Each entry: (ReqType, req_to_dict, RespType, dict_to_resp)

def _mk_req(msg) -> dict:
    return {"path": msg.path, "create_parents": msg.opts.create_parent,
            "mode": msg.opts.mode}


def _mk_resp(h: dict, msg) -> None:
    msg.flag = True
    _status_to_proto(h.get("status", {}), msg.status)


def _create_req(msg) -> dict:
    return {"path": msg.path,
            "block_size": msg.opts.block_size,
            "replicas": msg.opts.replicas,
            "storage_tier": _WIRE_TO_TIER.get(
                msg.opts.storage_policy.storage_type, ""),
            "overwrite": bool(msg.flags & _O_TRUNC),
            "mode": msg.opts.mode}


def _create_resp(h: dict, msg) -> None:
    _status_to_proto(h.get("status", {}), msg.file_status)


def _open_req(msg) -> dict:
    return {"path": msg.path}


def _open_resp(h: dict, msg) -> None:
    _file_blocks_to_proto(h.get("file_blocks", {}), msg.file_blocks)


def _delete_req(msg) -> dict:
    return {"path": msg.path, "recursive": msg.recursive}


def _delete_resp(h: dict, msg) -> None:
    msg.res.inodes = 1
    msg.res.bytes = 0


def _status_req(msg) -> dict:
    return {"path": msg.path}


def _status_resp(h: dict, msg) -> None:
    _status_to_proto(h.get("status", {}), msg.status)


def _exists_req(msg) -> dict:
    return {"path": msg.path}


def _exists_resp(h: dict, msg) -> None:
    msg.exists = bool(h.get("exists"))


def _list_req(msg) -> dict:
    return {"path": msg.path}


def _list_resp(h: dict, msg) -> None:
    for st in h.get("statuses", []):
        _status_to_proto(st, msg.statuses.add())


def _rename_req(msg) -> dict:
    return {"src": msg.src, "dst": msg.dst}


def _rename_resp(h: dict, msg) -> None:
    msg.result = True


def _add_block_req(msg) -> dict:
    out = {"path": msg.path,
           "exclude_workers": list(msg.exclude_workers),
           "client_host": msg.client_address.hostname,
           "commit_prev_len": -1}
    if msg.commit_blocks:
        out["commit_prev_len"] = msg.commit_blocks[-1].block_len
    return out


def _add_block_resp(h: dict, msg) -> None:
    b = h.get("block", {})
    msg.block.block.id = b.get("block", {}).get("block_id", 0)
    msg.block.block.block_size = b.get("block", {}).get("length", 0)
    tiers = b.get("tiers") or ["MEM"]
    msg.block.block.storage_type = _TIER_TO_WIRE.get(tiers[0], 0)
    msg.block.block.file_type = 1
    msg.block.offset = b.get("offset", 0)
    for a in b.get("locations", []):
        _addr_to_proto(a, msg.block.locs.add())


def _complete_req(msg) -> dict:
    commits = []
    block_lens = []
    for c in msg.commit_blocks:
        block_lens.append(c.block_len)
        commits.append({
            "block_id": c.block_id,
            "locations": [l.worker_id for l in c.locations],
            "tiers": [_WIRE_TO_TIER.get(l.storage_type, "MEM")
                      for l in c.locations]})
    return {"path": msg.path, "length": msg.len,
            "block_lens": block_lens or None, "commits": commits}


def _complete_resp(h: dict, msg) -> None:
    msg.result = True


def _block_locs_req(msg) -> dict:
    return {"path": msg.path}


def _block_locs_resp(h: dict, msg) -> None:
    _file_blocks_to_proto(h.get("file_blocks", {}), msg.blocks)


def _free_req(msg) -> dict:
    return {"path": msg.path, "recursive": msg.recursive}


def _free_resp(h: dict, msg) -> None:
    msg.res.inodes = 0
    msg.res.bytes = h.get("freed_blocks", 0)


def _bwrite_req(msg) -> dict:
    return {"block_id": msg.block.id, "reserve": msg.block_size,
            "tier": _WIRE_TO_TIER.get(msg.block.storage_type, "")}


def _bwrite_resp(h: dict, msg) -> None:
    msg.id = 0
    msg.off = 0
    msg.block_size = 0
    msg.storage_type = _TIER_TO_WIRE.get(h.get("tier", "MEM"), 0)


def _bread_req(msg) -> dict:
    return {"block_id": msg.id, "offset": msg.off, "length": msg.len,
            "chunk_size": msg.chunk_size}


def _bread_resp(h: dict, msg) -> None:
    msg.id = 0
    msg.len = h.get("length", 0)
    msg.storage_type = 0


def _ci_to_dict(ci) -> dict:
    return {"component": ci.component,
            "release_version": ci.release_version,
            "protocol_version": ci.protocol_version or 1}


def _ci_fill(msg, d: dict) -> None:
    msg.component = d.get("component", "")
    msg.release_version = d.get("release_version", "")
    msg.protocol_version = d.get("protocol_version", 1)


def _fsinfo_req(msg) -> dict:
    out: dict = {}
    if msg.HasField("component_info"):
        out["component_info"] = _ci_to_dict(msg.component_info)
    return out


def _fsinfo_resp(h: dict, msg) -> None:
    msg.active_master = h.get("cluster_id", "")
    msg.inode_dir_num = 0
    msg.inode_file_num = h.get("inode_num", 0)
    msg.block_num = h.get("block_num", 0)
    cap = h.get("capacity", 0)
    used = h.get("used", 0)
    msg.capacity = cap
    msg.available = max(0, cap - used)
    msg.fs_used = used
    msg.non_fs_used = 0
    msg.reserved_bytes = 0
    msg.allocatable_capacity = cap
    msg.allocatable_available = max(0, cap - used)
    for field_name in ("live_workers", "decommission_workers",
                       "lost_workers"):
        for w in h.get(field_name, []):
            _fill_worker(getattr(msg, field_name).add(), w)
    ci = h.get("component_info")
    if ci:
        # advertise the master's compatibility contract (handshake)
        _ci_fill(msg.compatibility.server, ci)
        msg.compatibility.compatibility_mode = 1   # DIAGNOSE


def _fill_worker(wp, w: dict) -> None:
    a = w.get("address", {})
    wp.address.worker_id = a.get("worker_id", 0)
    wp.address.hostname = a.get("hostname", "")
    wp.address.ip_addr = a.get("hostname", "")
    wp.address.rpc_port = a.get("rpc_port", 0)
    wp.address.web_port = 0
    wcap = wused = 0
    for st in w.get("storages", []):
        sp = wp.storage_map[f"{st.get('tier', 'MEM')}-"
                            f"{st.get('dir_id', 0)}"]
        sp.dir_id = st.get("dir_id", 0)
        sp.storage_id = f"{st.get('tier', 'MEM')}-{st.get('dir_id', 0)}"
        sp.failed = False
        sp.capacity = st.get("capacity", 0)
        sp.fs_used = st.get("used", 0)
        sp.available = max(0, sp.capacity - sp.fs_used)
        sp.non_fs_used = 0
        sp.reserved_bytes = 0
        sp.storage_type = _TIER_TO_WIRE.get(st.get("tier", "MEM"), 0)
        sp.block_num = st.get("block_num", 0)
        sp.dir_path = ""
        wcap += sp.capacity
        wused += sp.fs_used
    wp.capacity = wcap
    wp.available = max(0, wcap - wused)
    wp.fs_used = wused
    wp.non_fs_used = 0
    wp.reserved_bytes = 0
    wp.last_update = w.get("last_heartbeat_ms", 0)
    ci = w.get("component_info")
    if ci:
        _ci_fill(wp.component_info, ci)


CODECS: dict[int, tuple] = {
    int(RpcCode.Mkdir): (M["MkdirRequest"], _mk_req,
                         M["MkdirResponse"], _mk_resp),
    int(RpcCode.CreateFile): (M["CreateFileRequest"], _create_req,
                              M["CreateFileResponse"], _create_resp),
    int(RpcCode.OpenFile): (M["OpenFileRequest"], _open_req,
                            M["OpenFileResponse"], _open_resp),
    int(RpcCode.Delete): (M["DeleteRequest"], _delete_req,
                          M["DeleteResponse"], _delete_resp),
    int(RpcCode.FileStatus): (M["GetFileStatusRequest"], _status_req,
                              M["GetFileStatusResponse"], _status_resp),
    int(RpcCode.Exists): (M["ExistsRequest"], _exists_req,
                          M["ExistsResponse"], _exists_resp),
    int(RpcCode.ListStatus): (M["ListStatusRequest"], _list_req,
                              M["ListStatusResponse"], _list_resp),
    int(RpcCode.Rename): (M["RenameRequest"], _rename_req,
                          M["RenameResponse"], _rename_resp),
    int(RpcCode.AddBlock): (M["AddBlockRequest"], _add_block_req,
                            M["AddBlockResponse"], _add_block_resp),
    int(RpcCode.CompleteFile): (M["CompleteFileRequest"], _complete_req,
                                M["CompleteFileResponse"], _complete_resp),
    int(RpcCode.GetBlockLocations): (
        M["GetBlockLocationsRequest"], _block_locs_req,
        M["GetBlockLocationsResponse"], _block_locs_resp),
    int(RpcCode.Free): (M["FreeRequest"], _free_req,
                        M["FreeResponse"], _free_resp),
    int(RpcCode.GetFilesystemInfo): (
        M["GetFilesystemInfoRequest"], _fsinfo_req,
        M["GetFilesystemInfoResponse"], _fsinfo_resp),
    int(RpcCode.WriteBlock): (M["BlockWriteRequest"], _bwrite_req,
                              M["BlockWriteResponse"], _bwrite_resp),
    int(RpcCode.ReadBlock): (M["BlockReadRequest"], _bread_req,
                             M["BlockReadResponse"], _bread_resp),
}


# ---------------------------------------------------------- error frames
# The reference carries errors in the DATA section (rpc_message.rs:193-197
# error_ext), encoded by its ErrorEncoder (error_encoder.rs:24-50):
#   i32 kind BE | u32 msg_len BE | msg utf8 | u32 data_len BE | data
# with kind from curvine-error ErrorKind (fs_error.rs:38-79).  Our typed
# errors map onto those kinds; unmapped ones ride as Common (10000).

_ERROR_KINDS = {
    "NotLeader": 2, "RpcTimeout": 4,
    "FileAlreadyExists": 7, "FileNotFound": 8,
    "ParentNotDir": 10, "DirNotEmpty": 11,
    "ChecksumMismatch": 12, "BlockInWriting": 13,
    "InvalidPath": 16, "CapacityExceeded": 17,
    "Unsupported": 19, "UfsError": 20, "Expired": 21,
    "JobNotFound": 23, "IsDirectory": 26, "NotDirectory": 27,
    "InvalidArgument": 28, "BlockNotFound": 29,
    "NoAvailableWorker": 31,
}
_KIND_NAMES = {v: k for k, v in _ERROR_KINDS.items()}


def encode_error(e: Exception) -> bytes:
    import struct as _st
    kind = _ERROR_KINDS.get(type(e).__name__, 10000)
    msg = str(e).encode()
    return _st.pack(">i", kind) + _st.pack(">I", len(msg)) + msg + \
        _st.pack(">I", 0)


def decode_error(raw: bytes):
    """Parse a reference error blob -> (our typed exception)."""
    import struct as _st

    from curvine_amd import errors as _err
    if len(raw) < 8:
        return _err.FsError(raw.decode(errors="replace") or "rpc error")
    (kind,) = _st.unpack_from(">i", raw, 0)
    (n,) = _st.unpack_from(">I", raw, 4)
    msg = raw[8:8 + n].decode(errors="replace")
    name = _KIND_NAMES.get(kind)
    cls = getattr(_err, name, None) if name else None
    return (cls or _err.FsError)(msg)


def decode_request(code: int, raw: bytes) -> Optional[dict]:
    """Parse a protobuf request header into the internal dict, or None
    when the code has no protobuf codec / the bytes don't parse."""
    ent = CODECS.get(code)
    if ent is None:
        return None
    req_type, req_to_dict = ent[0], ent[1]
    msg = req_type()
    try:
        msg.MergeFromString(raw)
    except Exception:  # noqa: BLE001 — not protobuf
        return None
    return req_to_dict(msg)


def encode_response(code: int, header: dict) -> Optional[bytes]:
    ent = CODECS.get(code)
    if ent is None:
        return None
    resp_type, dict_to_resp = ent[2], ent[3]
    msg = resp_type()
    dict_to_resp(header or {}, msg)
    return msg.SerializeToString()


# client-direction helpers (tests + protobuf-speaking clients)

def encode_request(code: int, header: dict) -> Optional[bytes]:
    """Build a protobuf request from the internal dict (what a reference
    client would send; used by the golden/interop tests)."""
    ent = CODECS.get(code)
    if ent is None:
        return None
    req_type = ent[0]
    msg = req_type()
    h = header or {}
    c = RpcCode(code)
    if c == RpcCode.Mkdir:
        msg.path = h["path"]
        msg.opts.create_parent = bool(h.get("create_parents", True))
        msg.opts.mode = h.get("mode", 0o755)
        _policy(msg.opts.storage_policy, "MEM")
        msg.opts.owner = ""
        msg.opts.group = ""
    elif c in (RpcCode.CreateFile, RpcCode.OpenFile):
        msg.path = h["path"]
        o = msg.opts
        o.create_flag = 1
        o.create_parent = bool(h.get("create_parents", False))
        o.file_type = 1
        o.replicas = h.get("replicas", 1) or 1
        o.block_size = h.get("block_size", 0) or 134217728
        _policy(o.storage_policy, h.get("storage_tier") or "MEM")
        o.client_name = "curvine-amd"
        o.mode = h.get("mode", 0o644)
        o.owner = ""
        o.group = ""
        o.sync_ufs_meta = False
        o.ufs_len = 0
        msg.flags = (_O_CREAT | _O_TRUNC) if h.get("overwrite") else _O_CREAT
        if c == RpcCode.OpenFile:
            msg.flags = 0
    elif c in (RpcCode.FileStatus, RpcCode.GetBlockLocations):
        msg.path = h["path"]
    elif c == RpcCode.Exists:
        msg.path = h["path"]
    elif c == RpcCode.ListStatus:
        msg.path = h["path"]
        msg.need_location = False
    elif c in (RpcCode.Delete, RpcCode.Free):
        msg.path = h["path"]
        msg.recursive = bool(h.get("recursive", False))
    elif c == RpcCode.Rename:
        msg.src = h["src"]
        msg.dst = h["dst"]
        msg.flags = 0
    elif c == RpcCode.AddBlock:
        msg.path = h["path"]
        msg.located = True
        msg.client_address.client_name = "curvine-amd"
        msg.client_address.hostname = h.get("client_host", "")
        msg.client_address.ip_addr = h.get("client_host", "")
        msg.client_address.port = 0
        msg.file_len = 0
        for w in h.get("exclude_workers") or []:
            msg.exclude_workers.append(w)
    elif c == RpcCode.CompleteFile:
        msg.path = h["path"]
        msg.len = h.get("length", 0)
        msg.client_name = "curvine-amd"
        msg.only_flush = False
        for i, bl in enumerate(h.get("block_lens") or []):
            cb = msg.commit_blocks.add()
            commits = h.get("commits") or []
            cb.block_id = commits[i]["block_id"] if i < len(commits) else 0
            cb.block_len = bl
    elif c == RpcCode.WriteBlock:
        msg.block.id = h["block_id"]
        msg.block.block_size = h.get("reserve", 0)
        msg.block.storage_type = _TIER_TO_WIRE.get(h.get("tier") or "MEM", 0)
        msg.block.file_type = 1
        msg.off = 0
        msg.block_size = h.get("reserve", 0)
        msg.short_circuit = False
        msg.client_name = "curvine-amd"
        msg.chunk_size = 1 << 20
    elif c == RpcCode.ReadBlock:
        msg.id = h["block_id"]
        msg.off = h.get("offset", 0)
        msg.len = h.get("length", 0)
        msg.chunk_size = h.get("chunk_size", 1 << 20)
        msg.short_circuit = False
        msg.enable_read_ahead = True
        msg.read_ahead_len = 4194304
        msg.drop_cache_len = 1048576
    return msg.SerializeToString()


def decode_response(code: int, raw: bytes) -> Optional[dict]:
    ent = CODECS.get(code)
    if ent is None:
        return None
    resp_type = ent[2]
    msg = resp_type()
    try:
        msg.MergeFromString(raw)
    except Exception:  # noqa: BLE001
        return None
    c = RpcCode(code)
    if c == RpcCode.Mkdir:
        return {"status": _status_from_proto(msg.status)}
    if c == RpcCode.CreateFile:
        return {"status": _status_from_proto(msg.file_status)}
    if c == RpcCode.OpenFile:
        return {"file_blocks": _file_blocks_from_proto(msg.file_blocks)}
    if c == RpcCode.GetBlockLocations:
        return {"file_blocks": _file_blocks_from_proto(msg.blocks)}
    if c == RpcCode.FileStatus:
        return {"status": _status_from_proto(msg.status)}
    if c == RpcCode.Exists:
        return {"exists": msg.exists}
    if c == RpcCode.ListStatus:
        return {"statuses": [_status_from_proto(s) for s in msg.statuses]}
    if c == RpcCode.Rename:
        return {}
    if c == RpcCode.Delete:
        return {"deleted_blocks": 0}
    if c == RpcCode.Free:
        return {"freed_blocks": msg.res.bytes}
    if c == RpcCode.CompleteFile:
        return {"status": {}}
    if c == RpcCode.AddBlock:
        lb = msg.block
        return {"block": {
            "block": {"block_id": lb.block.id,
                      "length": lb.block.block_size, "state": 0},
            "offset": lb.offset,
            "locations": [_addr_from_proto(a) for a in lb.locs],
            "tiers": [_WIRE_TO_TIER.get(lb.block.storage_type, "MEM")]
                     * len(lb.locs)}}
    if c == RpcCode.WriteBlock:
        return {"tier": _WIRE_TO_TIER.get(msg.storage_type, "MEM")}
    if c == RpcCode.ReadBlock:
        return {"length": msg.len}
    return None
