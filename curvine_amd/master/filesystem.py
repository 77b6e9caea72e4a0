"""MasterFilesystem — the metadata API.

Analog of the reference's `MasterFilesystem`
(/root/reference/curvine-master/src/master/fs/master_filesystem.rs:224-1741:
mkdir/create/open/delete/rename/add_block/complete_file/
get_block_locations/set_attr/symlink/link/resize, worker choose :579-600,
block_report :1289, delete_locations :1581) wrapping FsDir + WorkerManager.

Runs inside the master's single asyncio loop: methods are synchronous and
therefore atomic with respect to each other (the analog of the reference's
RwLock'd SyncFsDir fast path).
"""
from __future__ import annotations

import logging

from curvine_amd import errors as err
from curvine_amd.conf import ClusterConf
from curvine_amd.master.fs_dir import FsDir, Inode, norm_path
from curvine_amd.master.journal import JournalLoader, JournalWriter
from curvine_amd.master.worker_manager import WorkerManager
from curvine_amd.model import (BlockInfo, BlockState, FileBlocks, FileStatus,
                               FileType, LocatedBlock, WorkerInfo, now_ms)

log = logging.getLogger("curvine.masterfs")


class MasterFilesystem:
    def __init__(self, conf: ClusterConf, journal: JournalWriter | None = None):
        self.conf = conf
        self.journal = journal or JournalWriter(
            conf.journal.journal_dir, conf.journal.segment_max_bytes)
        self.fs_dir = FsDir(self.journal)
        self.workers = WorkerManager(conf.master.worker_expire_ms)
        self.loader = JournalLoader(conf.journal.journal_dir)
        # files currently being written: inode_id -> writer lease info
        self.writing: dict[int, dict] = {}

    # ---------------- lifecycle ----------------
    def restore(self) -> int:
        """Replay snapshot + WAL on startup."""
        last = self.loader.load(self.fs_dir.apply_entry, self.fs_dir.load_snapshot)
        self.journal.op_id = max(self.journal.op_id, last)
        if last:
            log.info("journal replayed through op_id=%d (%d inodes)",
                     last, len(self.fs_dir.inodes))
        return last

    def checkpoint(self) -> None:
        self.loader.save_snapshot(self.fs_dir.to_snapshot())
        self.journal.purge_through(self.fs_dir.journal.op_id)

    # ---------------- namespace ops ----------------
    def mkdir(self, path: str, mode: int = 0o755, create_parents: bool = True) -> FileStatus:
        node = self.fs_dir.mkdir(path, mode, create_parents)
        return self.fs_dir.status_of(node)

    def _create_node(self, path: str, block_size: int, replicas: int,
                     storage_tier: str, overwrite: bool, mode: int) -> Inode:
        node, removed = self.fs_dir.create(
            path,
            block_size or self.conf.master.block_size,
            min(max(replicas or self.conf.master.min_replication, 1),
                self.conf.master.max_replication),
            storage_tier or self.conf.client.storage_tier,
            overwrite, mode)
        if removed:
            self.workers.schedule_block_delete(removed)
        self.writing[node.id] = {"since_ms": now_ms()}
        return node

    def create(self, path: str, block_size: int = 0, replicas: int = 0,
               storage_tier: str = "", overwrite: bool = False,
               mode: int = 0o644) -> FileStatus:
        node = self._create_node(path, block_size, replicas, storage_tier,
                                 overwrite, mode)
        return self.fs_dir.status_of(node, norm_path(path))

    def create_dict(self, path: str, block_size: int = 0, replicas: int = 0,
                    storage_tier: str = "", overwrite: bool = False,
                    mode: int = 0o644) -> dict:
        """Reply-dict variant (mutation-QPS hot path)."""
        path = norm_path(path)
        node = self._create_node(path, block_size, replicas, storage_tier,
                                 overwrite, mode)
        return self.fs_dir.status_dict(node, path)

    def append(self, path: str) -> FileBlocks:
        node = self.fs_dir.must_resolve(path)
        if node.is_dir:
            raise err.IsDirectory(path)
        if node.id in self.writing:
            raise err.FileInWriting(path)
        node.complete = False
        self.writing[node.id] = {"since_ms": now_ms()}
        return self._file_blocks(node, path)

    def open(self, path: str) -> FileBlocks:
        node = self.fs_dir.must_resolve(path)
        if node.is_dir:
            raise err.IsDirectory(path)
        node.atime_ms = now_ms()
        node.access_count += 1
        return self._file_blocks(node, path)

    def _file_blocks(self, node: Inode, path: str) -> FileBlocks:
        blocks = []
        off = 0
        for bid, blen in node.blocks:
            locs = self.workers.locations_of(bid)
            blocks.append(LocatedBlock(
                block=BlockInfo(block_id=bid, length=blen,
                                state=int(BlockState.FINALIZED)),
                offset=off,
                locations=[w.address for w, _ in locs],
                tiers=[t for _, t in locs]))
            off += blen
        return FileBlocks(status=self.fs_dir.status_of(node, norm_path(path)),
                          blocks=blocks)

    def add_block(self, path: str, commit_prev_len: int = -1,
                  client_host: str = "", client_worker_id: int = -1,
                  exclude_workers: list[int] | None = None) -> LocatedBlock:
        from curvine_amd.fault import fault_point
        fault_point("master.add_block")
        node = self.fs_dir.must_resolve(path)
        if node.id not in self.writing:
            raise err.FsError(f"no write lease on {path}")
        workers = self.workers.choose_workers(
            node.replicas, self.conf.master.worker_policy,
            client_host, client_worker_id, set(exclude_workers or []))
        bid = self.fs_dir.add_block(node, commit_prev_len)
        off = sum(b[1] for b in node.blocks[:-1])
        return LocatedBlock(
            block=BlockInfo(block_id=bid, length=0, state=int(BlockState.WRITING)),
            offset=off,
            locations=[w.address for w in workers],
            tiers=[node.storage_tier] * len(workers))

    def _complete_node(self, path: str, length: int,
                       block_lens: list[int] | None,
                       commits: list[dict] | None) -> Inode:
        node = self.fs_dir.must_resolve(path)
        self.fs_dir.complete_file(node, length, block_lens)
        self.writing.pop(node.id, None)
        # client-reported block locations (commit metadata,
        # block_writer.rs:324-337 analog) — soft state, ahead of heartbeats
        for c in commits or []:
            for wid, tier in zip(c.get("locations", []),
                                 c.get("tiers", []) or ["MEM"] * len(c.get("locations", []))):
                self.workers.add_location(c["block_id"], wid, tier)
        return node

    def complete_file(self, path: str, length: int,
                      block_lens: list[int] | None = None,
                      commits: list[dict] | None = None) -> FileStatus:
        node = self._complete_node(path, length, block_lens, commits)
        return self.fs_dir.status_of(node, norm_path(path))

    def complete_file_dict(self, path: str, length: int,
                           block_lens: list[int] | None = None,
                           commits: list[dict] | None = None) -> dict:
        node = self._complete_node(path, length, block_lens, commits)
        return self.fs_dir.status_dict(node, norm_path(path))

    def delete(self, path: str, recursive: bool = False) -> int:
        removed = self.fs_dir.delete(path, recursive)
        self.workers.schedule_block_delete(removed)
        return len(removed)

    def rename(self, src: str, dst: str) -> None:
        self.fs_dir.rename(src, dst)

    def file_status(self, path: str) -> FileStatus:
        node = self.fs_dir.must_resolve(path)
        return self.fs_dir.status_of(node, norm_path(path))

    def exists(self, path: str) -> bool:
        return self.fs_dir.resolve(path) is not None

    def list_status(self, path: str) -> list[FileStatus]:
        node = self.fs_dir.must_resolve(path)
        if not node.is_dir:
            return [self.fs_dir.status_of(node, norm_path(path))]
        base = norm_path(path).rstrip("/")
        out = []
        for name, cid in sorted(node.children.items()):
            child = self.fs_dir.inodes[cid]
            out.append(self.fs_dir.status_of(child, f"{base}/{name}"))
        return out

    def get_block_locations(self, path: str) -> FileBlocks:
        return self.open(path)

    def set_attr(self, path: str, **attrs) -> FileStatus:
        node = self.fs_dir.must_resolve(path)
        clean = {k: v for k, v in attrs.items() if v is not None}
        xattrs = clean.pop("xattrs", None)
        if clean:
            self.fs_dir.set_attr(node, **clean)
        if xattrs:
            for k, v in xattrs.items():
                if v is None:
                    self.fs_dir.remove_xattr(node, k)
                else:
                    self.fs_dir.set_xattr(node, k, v)
        return self.fs_dir.status_of(node, norm_path(path))

    def symlink(self, link_path: str, target: str) -> FileStatus:
        node = self.fs_dir.symlink(link_path, target)
        return self.fs_dir.status_of(node, norm_path(link_path))

    def link(self, src: str, dst: str) -> FileStatus:
        node = self.fs_dir.link(src, dst)
        return self.fs_dir.status_of(node, norm_path(dst))

    def resize(self, path: str, new_length: int) -> FileStatus:
        node = self.fs_dir.must_resolve(path)
        removed = self.fs_dir.resize(node, new_length)
        if removed:
            self.workers.schedule_block_delete(removed)
        return self.fs_dir.status_of(node, norm_path(path))

    def free(self, path: str, recursive: bool = False) -> int:
        node = self.fs_dir.must_resolve(path)
        targets = [node]
        if node.is_dir:
            if not recursive and node.children:
                raise err.DirNotEmpty(path)
            stack = [node]
            targets = []
            while stack:
                n = stack.pop()
                if n.is_dir and n.children:
                    stack.extend(self.fs_dir.inodes[c] for c in n.children.values())
                elif not n.is_dir:
                    targets.append(n)
        count = 0
        for n in targets:
            removed = self.fs_dir.free(n)
            self.workers.schedule_block_delete(removed)
            count += len(removed)
        return count

    # ---------------- worker plane ----------------
    def worker_heartbeat(self, info: WorkerInfo, added: list[dict],
                         removed: list[int]) -> list[dict]:
        # drop reports for blocks the master no longer knows
        valid_added, stale = [], []
        for b in added:
            if b["block_id"] in self.fs_dir.block_index:
                valid_added.append(b)
            else:
                stale.append(b["block_id"])
        cmds = self.workers.heartbeat(info, valid_added, removed)
        for bid in stale:
            cmds.append({"cmd": "delete_block", "block_id": bid})
        return cmds

    def block_report(self, worker_id: int, blocks: list[dict]) -> list[int]:
        valid, to_delete = [], []
        for b in blocks:
            if b["block_id"] in self.fs_dir.block_index:
                valid.append(b)
            else:
                to_delete.append(b["block_id"])
        self.workers.block_report(worker_id, valid)
        return to_delete

    def handle_lost_workers(self, lost: list[int]) -> list[int]:
        """Returns block ids now under-replicated."""
        affected: list[int] = []
        for wid in lost:
            affected.extend(self.workers.remove_worker_locations(wid))
        return [b for b in affected if b in self.fs_dir.block_index]

    # ---------------- info ----------------
    def master_info(self) -> dict:
        from curvine_amd.model import WorkerState
        live = self.workers.live_workers()
        by_state: dict[int, list] = {}
        for w in self.workers.workers.values():
            by_state.setdefault(int(w.state), []).append(w.to_dict())
        return {
            "cluster_id": self.conf.cluster_id,
            "inode_num": len(self.fs_dir.inodes),
            "block_num": len(self.fs_dir.block_index),
            "live_workers": [w.to_dict() for w in live],
            "decommission_workers":
                by_state.get(int(WorkerState.DECOMMISSIONING), []),
            "lost_workers": by_state.get(int(WorkerState.LOST), []),
            "capacity": self.workers.total_capacity(),
            "used": self.workers.total_used(),
            "block_size": self.conf.master.block_size,
        }
