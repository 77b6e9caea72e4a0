"""Master process composition.

Analog of /root/reference/curvine-master/src/master/master_server.rs
(:164-276): journal system -> filesystem -> mount/job/replication managers
-> background actor -> RPC server, in that start order.
"""
from __future__ import annotations

import asyncio
import logging
from typing import Optional

from curvine_amd.conf import ClusterConf
from curvine_amd.master.filesystem import MasterFilesystem
from curvine_amd.master.handler import MasterHandler
from curvine_amd.master.jobs import JobManager
from curvine_amd.master.journal import JournalWriter, Op
from curvine_amd.master.mount_manager import MountManager
from curvine_amd.master.replication import MasterReplicationManager
from curvine_amd.rpc.codes import RpcCode
from curvine_amd.rpc.server import HandlerService, RpcServer

log = logging.getLogger("curvine.master")

# codes a raft follower may serve.  Client READS also go to the leader
# (linearizable read-your-writes; followers may lag) — only the raft
# protocol itself and liveness probes are served by anyone.
_READONLY_OK = {
    RpcCode.Heartbeat, RpcCode.MetricsReport,
    RpcCode.RaftVote, RpcCode.RaftAppendEntries, RpcCode.RaftInstallSnapshot,
    RpcCode.RaftTransferLeader,
}
import curvine_amd.master.handler as _handler_mod
_handler_mod._READONLY_OK = _READONLY_OK


class RetryCache:
    """Bounded (cid, rid) -> reply cache for mutation dedup
    (fs/fs_retry_cache.rs analog)."""

    def __init__(self, size: int = 100_000, ttl_ms: int = 600_000):
        from collections import OrderedDict
        import time as _t
        self._d: "OrderedDict[tuple, tuple[float, dict]]" = OrderedDict()
        self.size = size
        self.ttl = ttl_ms / 1000.0
        self._now = _t.monotonic

    def get(self, key):
        ent = self._d.get(key)
        if ent is None:
            return None
        if self._now() - ent[0] > self.ttl:
            self._d.pop(key, None)
            return None
        return ent[1]

    def put(self, key, reply: dict) -> None:
        self._d[key] = (self._now(), reply)
        while len(self._d) > self.size:
            self._d.popitem(last=False)


class MasterService(HandlerService):
    def __init__(self, master: "Master"):
        self.master = master

    def get_message_handler(self):
        return MasterHandler(self.master)


class RaftJournalWriter(JournalWriter):
    """Journal writer whose durable log is the raft log: log() appends to
    the raft leader's log (index == op_id); the raft node replicates and
    the RPC handler withholds replies until commit."""

    def __init__(self, journal_dir: str):
        super().__init__(journal_dir)
        self.raft = None   # wired by Master after construction

    def log(self, op: str, **fields) -> dict:
        if not self.enabled:
            return {}
        # index is assigned deterministically before append so the entry is
        # serialized into the raft log with its final op_id; 'ts' is the
        # log-time stamp every replica replays identically (deterministic
        # create/mtime and TTL expiry across leader and followers)
        from curvine_amd.model import now_ms
        index = self.raft.log.last_index + 1
        entry = {"op": op, "op_id": index, "ts": now_ms(), **fields}
        got = self.raft.append_local(entry)
        assert got == index
        self.op_id = index
        if self.on_log is not None:
            self.on_log(entry)
        return entry

    def flush(self) -> None:
        if self.raft is not None:
            self.raft.log.flush()

    def close(self) -> None:
        self.flush()

    def purge_through(self, op_id: int) -> None:
        pass   # raft log compaction handles retention


class Master:
    def __init__(self, conf: ClusterConf):
        self.conf = conf
        self._raft_mode = len(conf.journal.peers) > 1
        if self._raft_mode:
            self.journal = RaftJournalWriter(conf.journal.journal_dir)
        else:
            self.journal = JournalWriter(conf.journal.journal_dir,
                                         conf.journal.segment_max_bytes)
        self.fs = MasterFilesystem(conf, self.journal)
        self.mounts = MountManager(self.journal)
        self.jobs = JobManager(self)
        self.replication = MasterReplicationManager(self.fs)
        self.raft = None
        self.retry_cache = RetryCache(conf.master.retry_cache_size,
                                      conf.master.retry_cache_ttl_ms)
        # recent journal entries for metadata delta paging (codes 28/29)
        import collections
        self.recent_entries = collections.deque(maxlen=20_000)
        self.journal.on_log = self.recent_entries.append
        self.rpc_service = MasterService(self)
        self.rpc = RpcServer("master", conf.master.hostname,
                             conf.master.rpc_port, self.rpc_service)
        self.native_meta = None   # NativeMetaFrontend when enabled
        self.inode_db = None      # SqliteInodeStore when enabled
        # client-pushed metrics snapshots (MetricsReport, code 60)
        self.client_metrics: dict[str, dict] = {}
        from curvine_amd.compat import CompatibilityPolicy, PeerWarnDedup
        self.compat_policy = CompatibilityPolicy.from_conf(
            conf.compatibility)
        self.compat_warn_workers = PeerWarnDedup("worker")
        self.compat_warn_clients = PeerWarnDedup("client")
        self._actor_task: Optional[asyncio.Task] = None
        self._stopped = asyncio.Event()
        self._mutation_count = 0
        self._transfer_conn = None

    def transfer_conn(self):
        """Pooled connection to the standalone transfer service."""
        if self._transfer_conn is None:
            from curvine_amd.rpc.client import ClusterConnector
            self._transfer_conn = ClusterConnector(
                [self.conf.job.service_addr],
                self.conf.client.rpc_timeout_ms)
        return self._transfer_conn

    def _make_raft(self):
        from curvine_amd.master.raft import RaftNode
        peers = {}
        for spec in self.conf.journal.peers:
            pid, _, addr = spec.partition("@")
            host, _, port = addr.rpartition(":")
            peers[int(pid)] = (host, int(port))
        return RaftNode(
            self.conf.journal.node_id, peers,
            self.conf.journal.journal_dir,
            apply_entry=self._apply_entry,
            make_snapshot=self._snapshot_state,
            load_snapshot=self._load_snapshot_state,
            rebuild=self._rebuild_state,
            election_timeout_ms=self.conf.journal.election_timeout_ms,
            heartbeat_ms=self.conf.journal.heartbeat_interval_ms,
            learners={int(x) for x in self.conf.journal.learners})

    def _apply_entry(self, e: dict) -> None:
        """Raft follower apply path (leader applied at append time).
        op_id advances BEFORE the apply so the commit-gated native mirror
        tags every mirror op with the entry actually being applied."""
        self.journal.op_id = max(self.journal.op_id, e["op_id"])
        if not self.mounts.apply_entry(e):
            self.fs.fs_dir.apply_entry(e)

    def _snapshot_state(self) -> dict:
        state = self.fs.fs_dir.to_snapshot()
        state["mounts"] = self.mounts.to_snapshot()
        return state

    def _load_snapshot_state(self, state: dict) -> None:
        self.fs.fs_dir.load_snapshot(state)
        self.mounts.load_snapshot(state.get("mounts", []))
        if self.native_meta is not None:   # whole-state swap: re-prime
            self.native_meta.attach()

    def _rebuild_state(self) -> None:
        """Re-derive the state machine from snapshot + committed raft log
        (called when stepping down with optimistic uncommitted applies)."""
        from curvine_amd.master.fs_dir import FsDir
        self.fs.fs_dir = FsDir(self.journal)
        self.mounts.mounts = {}
        snap = self.fs.loader.snapshot_path()
        import os as _os
        if _os.path.exists(snap):
            with open(snap, "rb") as f:
                from curvine_amd.master.journal import decode_stream
                entries = list(decode_stream(f))
            if entries:
                self._load_snapshot_state(entries[0])
        start = max(self.raft.log.snapshot_index, self.journal.op_id)
        for i in range(start + 1, self.raft.commit_index + 1):
            self._apply_entry(self.raft.log.entry_at(i))
        self.journal.op_id = self.raft.commit_index
        if self.native_meta is not None:   # fs_dir was replaced: re-prime
            self.native_meta.attach()

    # ---------------- lifecycle ----------------
    async def start(self) -> "Master":
        if self._raft_mode:
            self.raft = self._make_raft()
            self.journal.raft = self.raft
            # replay the local raft log ONLY up to the persisted commit
            # watermark: the uncommitted tail may be truncated by a new
            # leader, and entries applied at boot could then never be
            # rolled back (state divergence).  The tail is applied later:
            # by _apply_committed when a leader re-commits it, or by
            # _become_leader's backlog apply if WE win the election.
            lg = self.raft.log
            safe = self.raft.boot_commit
            for i in range(lg.snapshot_index + 1, safe + 1):
                self._apply_entry(lg.entry_at(i))
            self.raft.last_applied = safe
            self.journal.op_id = safe
        else:
            self._restore()
        if self.conf.master.native_meta:
            try:
                from curvine_amd.master.native_meta import NativeMetaFrontend
                self.native_meta = NativeMetaFrontend(
                    self, nthreads=self.conf.master.meta_threads)
                self.rpc = self.native_meta
            except Exception as e:  # noqa: BLE001 — CPU-only fallback
                log.warning("native meta frontend unavailable (%s); "
                            "using asyncio rpc server", e)
                self.native_meta = None
        await self.rpc.start()
        if self.inode_db is not None and self.native_meta is not None \
                and self.conf.master.max_resident_inodes > 0:
            self.inode_db.on_fault = self._native_fault_in
        self.conf.master.rpc_port = self.rpc.port
        if self.raft is not None:
            if self.native_meta is not None:
                self.raft.on_role_change = self.native_meta.set_serving
            self.raft.start()
        self._actor_task = asyncio.create_task(self._actor_loop())
        log.info("master started on %s:%d%s", self.conf.master.hostname,
                 self.rpc.port, " (raft)" if self.raft else "")
        return self

    async def stop(self) -> None:
        self._stopped.set()
        if self._actor_task:
            self._actor_task.cancel()
            try:
                await self._actor_task
            except (asyncio.CancelledError, Exception):  # noqa: BLE001
                pass
        if self.raft is not None:
            await self.raft.stop()
        await self.rpc.stop()
        if self.inode_db is not None:
            self.inode_db.flush(self.fs.fs_dir, self.mounts.to_snapshot(),
                                self.journal.op_id)
            self.inode_db.close()
        self.journal.close()

    def _restore(self) -> None:
        """Restore state, newest source first: sqlite inode store (one
        table scan), else snapshot; then replay the WAL tail."""
        def apply(e: dict) -> None:
            if not self.mounts.apply_entry(e):
                self.fs.fs_dir.apply_entry(e)
            self.journal.op_id = max(self.journal.op_id, e["op_id"])

        def load_snap(state: dict) -> int:
            op = self.fs.fs_dir.load_snapshot(state)
            self.mounts.load_snapshot(state.get("mounts", []))
            return op

        start = 0
        paged = self.conf.master.inode_db and \
            self.conf.master.max_resident_inodes > 0
        if self.conf.master.inode_db:
            import os as _os
            from curvine_amd.master.inode_db import SqliteInodeStore
            self.inode_db = SqliteInodeStore(
                _os.path.join(self.conf.journal.journal_dir, "inodes.db"))
            if paged:
                # beyond-RAM mode: restore watermarks + root only; attach
                # the mirror BEFORE the WAL tail so replayed mutations
                # mark their rows dirty instead of needing a full resync
                start = self.inode_db.load_paged(self.fs.fs_dir,
                                                 self.mounts) or 0
                self.inode_db.enable_paging(self.fs.fs_dir)
                self._attach_inode_db_mirror()
            else:
                start = self.inode_db.load(self.fs.fs_dir, self.mounts) or 0
        self.fs.loader.load(apply, load_snap, start_op=start)
        if self.inode_db is not None:
            fs_dir = self.fs.fs_dir
            if paged:
                if fs_dir.journal.op_id > start:
                    # a journal snapshot newer than the DB was loaded
                    # around the mirror: flush every resident inode
                    self.inode_db._dirty.update(dict.keys(fs_dir.inodes))
                return
            # non-paged: WAL-tail entries replayed above were applied
            # without the mirror, so mark everything once
            if fs_dir.journal.op_id > start:
                self.inode_db.resync(fs_dir)
            self._attach_inode_db_mirror()

    def _native_evict(self, iid: int) -> None:
        if self.native_meta is not None:
            self.native_meta.lib.meta_drop(self.native_meta.sid, iid)

    def _native_fault_in(self, node) -> None:
        """Paged-in inodes return to the C++ tree so repeat lookups are
        served natively again.  The node and its children edges go in
        under ONE tree lock — a concurrent native resolve must never see
        a directory with a partially filled children map (that would
        serve a spurious FileNotFound)."""
        nm = self.native_meta
        if nm is None:
            return
        import struct as _st

        from curvine_amd.master.native_meta import (_node_blob,
                                                    _pack_blocks)
        blob, n = _node_blob(node)
        parts = []
        if node.children:
            for name, cid in node.children.items():
                nb = name.encode()
                parts.append(_st.pack("<I", len(nb)) + nb +
                             _st.pack("<q", cid))
        nm.lib.meta_upsert_with_children(
            nm.sid, node.id, node.is_dir, blob, n, _pack_blocks(node),
            node.mtime_ms, b"".join(parts))

    def _attach_inode_db_mirror(self) -> None:
        fs_dir = self.fs.fs_dir
        cur = fs_dir.mirror
        if cur is None:
            fs_dir.mirror = self.inode_db
        elif cur is not self.inode_db:
            from curvine_amd.master.fs_dir import MirrorFanout
            ms = cur.mirrors if isinstance(cur, MirrorFanout) else [cur]
            if self.inode_db not in ms:
                fs_dir.mirror = MirrorFanout(ms + [self.inode_db])

    def checkpoint(self) -> None:
        if self.inode_db is not None and \
                self.conf.master.max_resident_inodes > 0:
            # paged mode: sqlite IS the checkpoint; a full-namespace
            # snapshot would defeat the memory bound.  Purge the journal
            # through the store's fully-flushed watermark only.
            row = self.inode_db.conn.execute(
                "SELECT v FROM meta WHERE k='op_id'").fetchone()
            if row is not None:
                self.journal.purge_through(
                    int.from_bytes(row[0], "little"))
            return
        state = self._snapshot_state()
        self.fs.loader.save_snapshot(state)
        if self.raft is not None:
            la = self.raft.last_applied
            self.raft.log.compact_to(la, self.raft.log.term_at(la))
        else:
            self.journal.purge_through(self.journal.op_id)

    # ---------------- background actor ----------------
    async def _actor_loop(self) -> None:
        """ScheduledExecutor analog (master/fs/master_actor.rs:30-149):
        heartbeat expiry, TTL cleanup, replication scan, quota/eviction,
        periodic checkpoint."""
        check_ms = self.conf.master.heartbeat_check_ms
        tick = 0
        while not self._stopped.is_set():
            try:
                await asyncio.sleep(check_ms / 1000.0)
            except asyncio.CancelledError:
                return
            tick += 1
            if self.raft is not None and not self.raft.is_leader:
                continue   # followers don't drive cluster mutations
            try:
                lost = self.fs.workers.check_expired()
                if lost:
                    for bid in self.fs.handle_lost_workers(lost):
                        self.replication.mark_under_replicated(bid)
                self.replication.check_all()
                self.replication.scan()
                if self.native_meta is not None:
                    # fold native open() counters into LFU/LRU inputs
                    self.native_meta.drain_access()
                self._ttl_sweep()
                self._eviction_sweep()
                if self.inode_db is not None:
                    # snapshot on this thread, commit on the store's
                    # writer thread: the tick never stalls mutations
                    # behind the sqlite transaction
                    self.inode_db.flush_async(self.fs.fs_dir,
                                              self.mounts.to_snapshot(),
                                              self.journal.op_id)
                    maxres = self.conf.master.max_resident_inodes
                    if maxres > 0:
                        self.inode_db.page_out(
                            self.fs.fs_dir, set(self.fs.writing),
                            maxres, self._native_evict)
                if self.journal.op_id and tick % 60 == 0:
                    self.checkpoint()
            except Exception as e:  # noqa: BLE001
                log.exception("master actor tick failed: %s", e)

    def _ttl_sweep(self) -> None:
        """TTL subsystem analog (meta/inode/ttl/ttl_manager.rs:35-49).
        Paged mode also consults the store's ttl_deadline column so
        paged-out inodes still expire."""
        from curvine_amd.model import now_ms
        now = now_ms()
        expired = []
        seen = set()
        for node in list(self.fs.fs_dir.inodes.values()):
            if node.ttl_ms > 0 and node.create_ms + node.ttl_ms < now \
                    and node.ttl_action in ("delete", "free"):
                expired.append(node)
                seen.add(node.id)
        if self.inode_db is not None and \
                self.conf.master.max_resident_inodes > 0:
            for iid in self.inode_db.ttl_expired_ids(now):
                if iid in seen:
                    continue
                node = self.fs.fs_dir.inodes.get(iid)   # faults in
                if node is not None and node.ttl_ms > 0 and \
                        node.create_ms + node.ttl_ms < now and \
                        node.ttl_action in ("delete", "free"):
                    expired.append(node)
        for node in expired:
            path = self.fs.fs_dir.path_of(node.id)
            try:
                if node.ttl_action == "delete":
                    self.fs.delete(path, recursive=True)
                else:
                    self.fs.free(path, recursive=True)
                log.info("ttl %s %s", node.ttl_action, path)
            except Exception:  # noqa: BLE001
                pass

    def _eviction_sweep(self) -> None:
        """Quota/eviction analog (quota/quota_manager.rs:31-83 +
        eviction/evictor.rs:36-107): when used capacity crosses the high
        watermark, free least-recently/least-frequently used complete
        files until below the low watermark."""
        policy = self.conf.master.eviction_policy
        if policy == "none":
            return
        cap = self.fs.workers.total_capacity()
        used = self.fs.workers.total_used()
        if cap <= 0 or used < cap * self.conf.master.eviction_high_watermark:
            return
        target = cap * self.conf.master.eviction_low_watermark
        files = [n for n in self.fs.fs_dir.iter_files()
                 if n.complete and n.blocks and n.id not in self.fs.writing]
        if self.inode_db is not None and \
                self.conf.master.max_resident_inodes > 0:
            # paged mode: the coldest files may not be resident
            res = {n.id for n in files}
            for iid in self.inode_db.cold_file_ids():
                if iid in res:
                    continue
                n = self.fs.fs_dir.inodes.get(iid)   # faults in
                if n is not None and n.complete and n.blocks and \
                        n.id not in self.fs.writing:
                    files.append(n)
        if policy == "lfu":
            files.sort(key=lambda n: (n.access_count, n.atime_ms))
        else:
            files.sort(key=lambda n: n.atime_ms)
        freed = 0
        for node in files:
            if used - freed <= target:
                break
            freed += sum(b[1] for b in node.blocks)
            path = self.fs.fs_dir.path_of(node.id)
            try:
                self.fs.free(path)
                log.info("evicted (freed) %s", path)
            except Exception:  # noqa: BLE001
                pass
