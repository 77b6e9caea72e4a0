"""POD domain types.

Analog of the reference's `curvine-model` crate
(/root/reference/crates/common/curvine-model: FileStatus, BlockInfo,
ExtendedBlock, LocatedBlock, WorkerAddress/Info/Status, StorageInfo...).
All types round-trip through msgpack-friendly dicts for the RPC layer.
"""
from __future__ import annotations

import time
from dataclasses import dataclass, field, asdict
from enum import IntEnum

from curvine_amd.conf import TIER_MEM


def now_ms() -> int:
    return int(time.time() * 1000)


class FileType(IntEnum):
    FILE = 0
    DIR = 1
    SYMLINK = 2


class BlockState(IntEnum):
    WRITING = 0
    FINALIZED = 1


class WorkerState(IntEnum):
    LIVE = 0
    DECOMMISSIONING = 1
    DECOMMISSIONED = 2
    LOST = 3


@dataclass
class WorkerAddress:
    worker_id: int = 0
    hostname: str = "127.0.0.1"
    rpc_port: int = 0
    # MI355X: GPU ordinal whose HBM arena this worker owns (-1 = none)
    device_id: int = -1

    def key(self) -> str:
        return f"{self.hostname}:{self.rpc_port}"

    def to_dict(self) -> dict:
        return asdict(self)

    @staticmethod
    def from_dict(d: dict) -> "WorkerAddress":
        return WorkerAddress(**d)


@dataclass
class StorageInfo:
    tier: str = TIER_MEM
    dir_id: int = 0
    capacity: int = 0
    used: int = 0
    block_num: int = 0

    @property
    def available(self) -> int:
        return max(0, self.capacity - self.used)


@dataclass
class WorkerInfo:
    address: WorkerAddress = field(default_factory=WorkerAddress)
    state: int = int(WorkerState.LIVE)
    storages: list[StorageInfo] = field(default_factory=list)
    last_heartbeat_ms: int = 0
    # structured version report (ComponentInfoProto analog, wire field
    # 1000); None = legacy peer, treated as unknown by the policy
    component_info: dict | None = None

    @property
    def capacity(self) -> int:
        return sum(s.capacity for s in self.storages)

    @property
    def used(self) -> int:
        return sum(s.used for s in self.storages)

    @property
    def available(self) -> int:
        return sum(s.available for s in self.storages)

    def to_dict(self) -> dict:
        d = asdict(self)
        return d

    @staticmethod
    def from_dict(d: dict) -> "WorkerInfo":
        w = WorkerInfo()
        w.address = WorkerAddress.from_dict(d["address"])
        w.state = d.get("state", 0)
        w.storages = [StorageInfo(**s) for s in d.get("storages", [])]
        w.last_heartbeat_ms = d.get("last_heartbeat_ms", 0)
        w.component_info = d.get("component_info")
        return w


@dataclass
class BlockInfo:
    """A block of a file: fixed id, length known after finalize."""
    block_id: int = 0
    length: int = 0
    state: int = int(BlockState.FINALIZED)

    def to_dict(self) -> dict:
        return asdict(self)


@dataclass
class LocatedBlock:
    block: BlockInfo = field(default_factory=BlockInfo)
    offset: int = 0                     # byte offset within the file
    locations: list[WorkerAddress] = field(default_factory=list)
    tiers: list[str] = field(default_factory=list)   # tier per location

    def to_dict(self) -> dict:
        return {
            "block": self.block.to_dict(),
            "offset": self.offset,
            "locations": [x.to_dict() for x in self.locations],
            "tiers": list(self.tiers),
        }

    @staticmethod
    def from_dict(d: dict) -> "LocatedBlock":
        return LocatedBlock(
            block=BlockInfo(**d["block"]),
            offset=d.get("offset", 0),
            locations=[WorkerAddress.from_dict(x) for x in d.get("locations", [])],
            tiers=list(d.get("tiers", [])),
        )


@dataclass
class FileStatus:
    inode_id: int = 0
    path: str = "/"
    name: str = ""
    file_type: int = int(FileType.FILE)
    length: int = 0
    is_complete: bool = True
    block_size: int = 64 << 20
    replicas: int = 1
    storage_tier: str = TIER_MEM
    mtime_ms: int = 0
    atime_ms: int = 0
    mode: int = 0o644
    uid: int = 0
    gid: int = 0
    ttl_ms: int = 0
    ttl_action: str = "none"     # none | delete | free
    symlink_target: str = ""
    nlink: int = 1
    xattrs: dict = field(default_factory=dict)

    @property
    def is_dir(self) -> bool:
        return self.file_type == FileType.DIR

    @property
    def is_symlink(self) -> bool:
        return self.file_type == FileType.SYMLINK

    # manual dict (dataclasses.asdict deep-copies: ~10x slower on the
    # metadata QPS hot path)
    def to_dict(self) -> dict:
        return {
            "inode_id": self.inode_id, "path": self.path, "name": self.name,
            "file_type": self.file_type, "length": self.length,
            "is_complete": self.is_complete, "block_size": self.block_size,
            "replicas": self.replicas, "storage_tier": self.storage_tier,
            "mtime_ms": self.mtime_ms, "atime_ms": self.atime_ms,
            "mode": self.mode, "uid": self.uid, "gid": self.gid,
            "ttl_ms": self.ttl_ms, "ttl_action": self.ttl_action,
            "symlink_target": self.symlink_target, "nlink": self.nlink,
            "xattrs": self.xattrs,
        }

    @staticmethod
    def from_dict(d: dict) -> "FileStatus":
        return FileStatus(**d)


@dataclass
class FileBlocks:
    """Open-file view: status + all located blocks (open RPC reply)."""
    status: FileStatus = field(default_factory=FileStatus)
    blocks: list[LocatedBlock] = field(default_factory=list)

    def to_dict(self) -> dict:
        return {"status": self.status.to_dict(),
                "blocks": [b.to_dict() for b in self.blocks]}

    @staticmethod
    def from_dict(d: dict) -> "FileBlocks":
        return FileBlocks(
            status=FileStatus.from_dict(d["status"]),
            blocks=[LocatedBlock.from_dict(b) for b in d.get("blocks", [])],
        )


@dataclass
class MountInfo:
    mount_id: int = 0
    curvine_path: str = ""
    ufs_path: str = ""
    properties: dict = field(default_factory=dict)
    # consistency: cache (cv is authority) | fs (ufs is authority)
    cache_mode: str = "cache"
    auto_cache: bool = True

    def to_dict(self) -> dict:
        return asdict(self)

    @staticmethod
    def from_dict(d: dict) -> "MountInfo":
        return MountInfo(**d)


@dataclass
class MasterInfo:
    cluster_id: str = ""
    leader: str = ""
    live_workers: list[dict] = field(default_factory=list)
    lost_workers: list[str] = field(default_factory=list)
    capacity: int = 0
    used: int = 0
    inode_num: int = 0
    block_num: int = 0


# Worker commands returned on heartbeat (analog of WorkerCommand)
CMD_DELETE_BLOCK = "delete_block"
CMD_REPLICATE = "replicate"
