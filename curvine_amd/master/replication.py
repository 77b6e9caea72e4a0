"""Master-side replication manager.

Analog of /root/reference/curvine-master/src/master/replication/
master_replication_manager.rs: tracks under-replicated blocks and submits
replication jobs to a source worker (delivered as heartbeat commands); the
worker pushes the block to the target and reports back
(SubmitBlockReplicationJob / ReportBlockReplicationResult codes).
"""
from __future__ import annotations

import logging

from curvine_amd.model import CMD_REPLICATE

log = logging.getLogger("curvine.replication")


class MasterReplicationManager:
    def __init__(self, fs):
        self.fs = fs
        self.pending: dict[int, dict] = {}   # block_id -> job
        self.next_job_id = 0

    def mark_under_replicated(self, block_id: int) -> None:
        if block_id in self.pending:
            return
        if block_id not in self.fs.fs_dir.block_index:
            return
        self.pending[block_id] = {"state": "queued"}

    def scan(self) -> int:
        """Find blocks with fewer live replicas than their file requires and
        queue replication. Returns number of jobs submitted."""
        submitted = 0
        for bid, job in list(self.pending.items()):
            if job["state"] != "queued":
                continue
            submitted += self._submit(bid, job)
        return submitted

    def check_all(self) -> None:
        """Periodic sweep: compare live locations vs required replicas."""
        from curvine_amd.model import WorkerState
        decom_exists = any(w.state == WorkerState.DECOMMISSIONING
                           for w in self.fs.workers.workers.values())
        for node in self.fs.fs_dir.iter_files():
            if not node.complete or (node.replicas <= 1 and not decom_exists):
                continue
            for bid, _ in node.blocks:
                locs = self.fs.workers.locations_of(bid)
                healthy = [1 for w, _ in locs
                           if w.state != WorkerState.DECOMMISSIONING]
                if 0 < len(locs) and len(healthy) < node.replicas:
                    self.mark_under_replicated(bid)

    def _submit(self, block_id: int, job: dict) -> int:
        inode_id = self.fs.fs_dir.block_index.get(block_id)
        if inode_id is None:
            self.pending.pop(block_id, None)
            return 0
        node = self.fs.fs_dir.inodes.get(inode_id)
        locs = self.fs.workers.locations_of(block_id)
        if node is None or not locs:
            self.pending.pop(block_id, None)
            return 0
        # replicas on decommissioning workers don't count toward the goal
        from curvine_amd.model import WorkerState
        healthy = [(w, t) for w, t in locs
                   if w.state != WorkerState.DECOMMISSIONING]
        if len(healthy) >= node.replicas:
            self.pending.pop(block_id, None)
            return 0
        have = {w.address.worker_id for w, _ in locs}
        try:
            targets = self.fs.workers.choose_workers(
                node.replicas - len(healthy), "load_based", exclude=have)
        except Exception:  # noqa: BLE001 — not enough workers yet
            return 0
        src = locs[0][0]
        blen = next((b[1] for b in node.blocks if b[0] == block_id), 0)
        self.next_job_id += 1
        job.update(state="running", job_id=self.next_job_id,
                   src=src.address.worker_id,
                   targets=[t.address.worker_id for t in targets])
        self.fs.workers.add_command(src.address.worker_id, {
            "cmd": CMD_REPLICATE, "job_id": self.next_job_id,
            "block_id": block_id, "block_len": blen,
            "tier": node.storage_tier,
            "targets": [t.address.to_dict() for t in targets]})
        log.info("replication job %d: block %d %s -> %s",
                 self.next_job_id, block_id, src.address.key(),
                 [t.address.key() for t in targets])
        return 1

    def report_result(self, h: dict) -> None:
        bid = h.get("block_id")
        if h.get("success"):
            self.pending.pop(bid, None)
        else:
            job = self.pending.get(bid)
            if job is not None:
                job["state"] = "queued"   # retry on next scan
            log.warning("replication of block %s failed: %s", bid, h.get("error"))
