"""Cluster configuration.

Single-TOML configuration tree mirroring the reference's `curvine-config`
crate (/root/reference/crates/common/curvine-config: `ClusterConf` root with
`MasterConf`/`WorkerConf`/`ClientConf`/`FuseConf`/`JournalConf`..., defaults
in etc/curvine-cluster.toml). Data-dir entries use the same
``"[MEM:30GB]/path"`` syntax (worker_conf.rs:26), extended with the MI355X
``HBM`` tier: ``"[HBM:200GB]/gpu0"`` (path component is a label; HBM blocks
live in a hipMalloc arena, not a filesystem).
"""
from __future__ import annotations

import dataclasses
import os
import re
from dataclasses import dataclass, field
from typing import Any

try:
    import tomli as _toml
except ImportError:  # pragma: no cover
    _toml = None

_UNITS = {
    "B": 1, "KB": 1 << 10, "MB": 1 << 20, "GB": 1 << 30, "TB": 1 << 40,
    "K": 1 << 10, "M": 1 << 20, "G": 1 << 30, "T": 1 << 40,
    "KIB": 1 << 10, "MIB": 1 << 20, "GIB": 1 << 30, "TIB": 1 << 40,
}


def parse_bytes(s: "str | int | float") -> int:
    """'64MB' → 67108864. Accepts int passthrough."""
    if isinstance(s, (int, float)):
        return int(s)
    m = re.fullmatch(r"\s*([0-9.]+)\s*([A-Za-z]*)\s*", s)
    if not m:
        raise ValueError(f"bad byte size: {s!r}")
    val = float(m.group(1))
    unit = m.group(2).upper() or "B"
    if unit not in _UNITS:
        raise ValueError(f"bad byte unit: {s!r}")
    return int(val * _UNITS[unit])


def fmt_bytes(n: float) -> str:
    for unit, div in (("TiB", 1 << 40), ("GiB", 1 << 30), ("MiB", 1 << 20), ("KiB", 1 << 10)):
        if abs(n) >= div:
            return f"{n / div:.2f} {unit}"
    return f"{int(n)} B"


# storage tiers, hottest first.  HBM is the MI355X addition (288 GB/GPU).
TIER_HBM = "HBM"
TIER_MEM = "MEM"
TIER_SSD = "SSD"
TIER_HDD = "HDD"
TIERS = (TIER_HBM, TIER_MEM, TIER_SSD, TIER_HDD)
TIER_ORDER = {t: i for i, t in enumerate(TIERS)}

_DATA_DIR_RE = re.compile(r"^\[(\w+)(?::([^\]]+))?\](.*)$")


@dataclass
class DataDir:
    tier: str = TIER_MEM
    capacity: int = 1 << 30
    path: str = "/tmp/curvine/data"
    device_id: int = 0  # GPU ordinal for HBM dirs
    # O_DIRECT reads on SSD/HDD dirs (NVMe page-cache bypass; the
    # reference's SPDK-tier analog at the kernel-API level):
    # "[SSD:1TB:direct]/nvme0"
    o_direct: bool = False

    @staticmethod
    def parse(s: str) -> "DataDir":
        """Parse "[MEM:30GB]/path" / "[HBM:200GB:0]label" /
        "[SSD:1TB:direct]/nvme" / bare path."""
        m = _DATA_DIR_RE.match(s)
        if not m:
            return DataDir(tier=TIER_SSD, capacity=0, path=s)
        tier = m.group(1).upper()
        if tier not in TIERS:
            raise ValueError(f"unknown storage tier in {s!r}")
        cap, dev, direct = 1 << 30, 0, False
        if m.group(2):
            parts = m.group(2).split(":")
            cap = parse_bytes(parts[0])
            for tok in parts[1:]:
                if tok.lower() == "direct":
                    direct = True
                elif tok:
                    dev = int(tok)
        return DataDir(tier=tier, capacity=cap,
                       path=m.group(3) or f"/tmp/curvine/{tier.lower()}",
                       device_id=dev, o_direct=direct)


@dataclass
class MasterConf:
    hostname: str = "127.0.0.1"
    rpc_port: int = 8995
    web_port: int = 9000
    meta_dir: str = "/tmp/curvine/meta"
    # worker placement policy: local | round_robin | random | load_based
    worker_policy: str = "local"
    min_replication: int = 1
    max_replication: int = 16
    block_size: int = 64 << 20
    heartbeat_check_ms: int = 5_000
    worker_expire_ms: int = 60_000
    retry_cache_size: int = 100_000
    retry_cache_ttl_ms: int = 600_000
    audit_log: bool = False
    # capacity watermarks for quota/eviction (fractions of cluster capacity)
    eviction_high_watermark: float = 0.95
    eviction_low_watermark: float = 0.85
    eviction_policy: str = "lru"  # lru | lfu | none
    ttl_check_ms: int = 5_000
    # native C++ metadata frontend (csrc/meta_server.cpp): epoll threads
    # serve FileStatus/ListStatus/Exists GIL-free, everything else is
    # forwarded to the Python handler.  Falls back to the asyncio server
    # if the native extension is unavailable.
    native_meta: bool = True
    meta_threads: int = 4
    # sqlite-backed durable inode store (RocksInodeStore analog): dirty
    # inodes batched per actor tick; restart = table scan + WAL tail.
    # Non-raft masters only (raft nodes rebuild from the raft log).
    inode_db: bool = True
    # beyond-RAM namespace: cap the resident Python/native inode maps and
    # page cold, flushed inodes to sqlite (0 = unlimited, paging off)
    max_resident_inodes: int = 0


@dataclass
class JournalConf:
    enable: bool = True
    journal_dir: str = "/tmp/curvine/journal"
    # raft peers "id@host:port" (port = the master RPC port); single entry
    # or empty = standalone (no election)
    peers: list[str] = field(default_factory=list)
    node_id: int = 0          # this master's id within `peers`
    snapshot_interval_entries: int = 100_000
    segment_max_bytes: int = 256 << 20
    flush_batch: int = 256
    flush_interval_ms: int = 10
    rpc_port: int = 8996
    election_timeout_ms: int = 1500
    heartbeat_interval_ms: int = 300
    # non-voting member ids: replicated to like any peer but excluded
    # from elections and the commit quorum (raft learners — warm
    # standbys / read replicas)
    learners: list[int] = field(default_factory=list)


@dataclass
class WorkerConf:
    hostname: str = "127.0.0.1"
    rpc_port: int = 8997
    web_port: int = 9001
    data_dirs: list[str] = field(default_factory=lambda: ["[MEM:1GB]/tmp/curvine/mem"])
    heartbeat_interval_ms: int = 3_000
    io_slow_us: int = 300_000
    # device staging: pinned ring buffers for HBM<->host movement
    staging_buf_bytes: int = 4 << 20
    staging_buf_count: int = 8
    replication_concurrency: int = 4
    # fsync block files at finalize.  Off by default: this is a CACHE —
    # a crash loses re-fetchable blocks, and the publish-time CRC catches
    # torn files at read/verify time.  Turn on for cache-as-of-record
    # deployments.
    fsync_on_finalize: bool = False
    # native C++ epoll data plane (csrc/data_server.cpp); falls back to
    # the asyncio RpcServer when the extension is unavailable
    native_data: bool = True
    data_threads: int = 12

    def parsed_dirs(self) -> list[DataDir]:
        return [DataDir.parse(s) for s in self.data_dirs]


@dataclass
class ClientConf:
    master_addrs: list[str] = field(default_factory=lambda: ["127.0.0.1:8995"])
    short_circuit: bool = True
    write_chunk_size: int = 1 << 20
    read_chunk_size: int = 1 << 20
    read_chunk_num: int = 8
    read_parallel: int = 1
    read_slice_size: int = 16 << 20
    write_buffer_chunks: int = 8
    replicas: int = 1
    block_size: int = 64 << 20
    rpc_timeout_ms: int = 60_000
    conn_retry: int = 3
    storage_tier: str = TIER_HBM  # preferred tier for new blocks
    enable_crc: bool = False
    # auto-cache UFS files on miss (unified fs)
    auto_cache: bool = True
    auto_cache_max_inflight: int = 4
    # client-side audit stream (unified_filesystem.rs:144-169 analog):
    # cmd/path/ok/used_us per metadata RPC on logger "audit.client"
    audit_log: bool = False


@dataclass
class FuseConf:
    mnt_path: str = "/tmp/curvine-fuse"
    mnt_number: int = 1          # parallel fuse channels (clone_fd analog)
    io_threads: int = 4
    max_write: int = 1 << 20
    max_readahead: int = 8 << 20
    attr_ttl_ms: int = 1_000
    entry_ttl_ms: int = 1_000
    negative_ttl_ms: int = 0
    direct_io: bool = False
    allow_other: bool = True
    state_file: str = "/tmp/curvine/fuse.state"
    kernel_cache: bool = True
    native_loop: bool = True   # GIL-free C++ READ channels
    # kernel writeback cache: dirty pages aggregate in the page cache and
    # arrive as large async WRITEs instead of one synchronous round trip
    # per write(2).  Off by default: the kernel raises internal EIO on
    # some unlinked-file page flush paths with no daemon op erroring
    # (see ROADMAP); the daemon-side semantics it needs (reads on
    # write-only handles, handle-served getattr, open-time block
    # pinning) are implemented and kept on
    # kernel page-cache write aggregation (FUSE_WRITEBACK_CACHE).  On by
    # default: writes ride the serialized Python path in this mode (the
    # native CAS-append window assumes in-order single-stream WRITEs)
    # and every FUSE suite + the fio byte-verification sweep passes.
    writeback_cache: bool = True
    # push per-op FUSE stats to the master every N seconds via
    # MetricsReport (0 = off)
    metrics_report_s: int = 30


@dataclass
class UfsConf:
    endpoint: str = ""
    access_key: str = ""
    secret_key: str = ""
    region: str = ""
    extra: dict = field(default_factory=dict)


@dataclass
class JobConf:
    worker_task_concurrency: int = 4
    task_chunk_size: int = 8 << 20
    store: str = "memory"  # memory | sqlite
    store_path: str = "/tmp/curvine/jobs.db"
    # standalone transfer service "host:port" ("" = jobs run embedded in
    # the master, the reference's other deployment shape)
    service_addr: str = ""


@dataclass
class CompatibilityConf:
    """Peer version policy (compatibility_conf.rs analog).  Defaults are
    intentionally lenient: diagnose mode with no bounds, so old
    components are never rejected without explicit configuration."""
    mode: str = "diagnose"          # diagnose | enforce
    min_worker_version: str = ""    # lowest worker release accepted
    min_client_version: str = ""    # lowest client release accepted
    blocked_versions: list = field(default_factory=list)


@dataclass
class ClusterConf:
    cluster_id: str = "curvine-amd"
    master: MasterConf = field(default_factory=MasterConf)
    journal: JournalConf = field(default_factory=JournalConf)
    worker: WorkerConf = field(default_factory=WorkerConf)
    client: ClientConf = field(default_factory=ClientConf)
    fuse: FuseConf = field(default_factory=FuseConf)
    ufs: UfsConf = field(default_factory=UfsConf)
    job: JobConf = field(default_factory=JobConf)
    compatibility: CompatibilityConf = field(
        default_factory=CompatibilityConf)
    testing: bool = False

    # ---------------- loading ----------------
    @staticmethod
    def from_dict(d: dict) -> "ClusterConf":
        conf = ClusterConf()
        for section, value in d.items():
            if not hasattr(conf, section):
                continue
            cur = getattr(conf, section)
            if dataclasses.is_dataclass(cur) and isinstance(value, dict):
                for k, v in value.items():
                    if hasattr(cur, k):
                        setattr(cur, k, v)
            else:
                setattr(conf, section, value)
        conf.apply_env()
        return conf

    @staticmethod
    def from_file(path: str) -> "ClusterConf":
        if _toml is None:
            raise RuntimeError("tomli not available")
        with open(path, "rb") as f:
            return ClusterConf.from_dict(_toml.load(f))

    def apply_env(self) -> None:
        """CURVINE_MASTER_HOSTNAME etc. overrides (cluster_conf.rs analog)."""
        env = os.environ
        if "CURVINE_MASTER_HOSTNAME" in env:
            self.master.hostname = env["CURVINE_MASTER_HOSTNAME"]
        if "CURVINE_WORKER_HOSTNAME" in env:
            self.worker.hostname = env["CURVINE_WORKER_HOSTNAME"]
        if "CURVINE_MASTER_PORT" in env:
            self.master.rpc_port = int(env["CURVINE_MASTER_PORT"])

    def master_addr(self) -> tuple[str, int]:
        return (self.master.hostname, self.master.rpc_port)

    def overlay(self, **kv: Any) -> "ClusterConf":
        """CLI overlay (ClientCliArgs analog): dotted keys, e.g.
        overlay(**{"client.replicas": 2}).  String values are coerced to
        the field's existing type (mount -o options arrive as text)."""
        for key, v in kv.items():
            obj = self
            parts = key.split("__") if "__" in key else key.split(".")
            for p in parts[:-1]:
                obj = getattr(obj, p)
            cur = getattr(obj, parts[-1], None)
            if isinstance(v, str) and not isinstance(cur, str):
                if isinstance(cur, bool):
                    v = v.lower() in ("1", "true", "yes", "on", "")
                elif isinstance(cur, int):
                    v = parse_bytes(v)
                elif isinstance(cur, float):
                    v = float(v)
            setattr(obj, parts[-1], v)
        return self
