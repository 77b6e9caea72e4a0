"""Worker registry, block-location map and placement policies.

Analog of the reference's `WorkerManager` + `BlockMap`/`WorkerMap`
(/root/reference/curvine-master/src/master/fs/worker_manager.rs,
fs/state/) and the `WorkerPolicy` placement strategies (fs/policy/*.rs:
local / round-robin / random / load-based, selected by
`master.worker_policy`).

Block locations are soft state: rebuilt from worker heartbeats/block
reports, never journaled (same as the reference).
"""
from __future__ import annotations

import logging
import random
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.conf import TIER_ORDER
from curvine_amd.model import (CMD_DELETE_BLOCK,
                               WorkerInfo, WorkerState, now_ms)

log = logging.getLogger("curvine.workers")


class WorkerManager:
    def __init__(self, expire_ms: int = 60_000):
        self.workers: dict[int, WorkerInfo] = {}
        self.expire_ms = expire_ms
        # block_id -> {worker_id: tier}
        self.block_locs: dict[int, dict[int, str]] = {}
        # pending commands per worker (delivered on heartbeat)
        self.commands: dict[int, list[dict]] = {}
        self._rr = 0
        # native metadata mirror (master/native_meta.WorkerMirror): keeps
        # the C++ OpenFile reply path's location map coherent
        self.mirror = None

    # ---------------- registry ----------------
    def heartbeat(self, info: WorkerInfo,
                  added_blocks: list[dict] | None = None,
                  removed_blocks: list[int] | None = None) -> list[dict]:
        wid = info.address.worker_id
        info.last_heartbeat_ms = now_ms()
        prev = self.workers.get(wid)
        if prev is not None and prev.state == WorkerState.DECOMMISSIONING:
            info.state = prev.state
        self.workers[wid] = info
        if self.mirror:
            self.mirror.upsert_worker(info)
        for b in added_blocks or []:
            self.add_location(b["block_id"], wid, b.get("tier", "MEM"))
        for bid in removed_blocks or []:
            locs = self.block_locs.get(bid)
            if locs:
                locs.pop(wid, None)
                if not locs:
                    self.block_locs.pop(bid, None)
                if self.mirror:
                    self.mirror.remove_loc(bid, wid)
        return self.commands.pop(wid, [])

    def add_location(self, block_id: int, worker_id: int, tier: str) -> None:
        """Single entry point for location inserts (heartbeat deltas,
        block reports, client commit metadata) so the native mirror sees
        every one."""
        self.block_locs.setdefault(block_id, {})[worker_id] = tier
        if self.mirror:
            self.mirror.add_loc(block_id, worker_id, tier)

    def block_report(self, worker_id: int, blocks: list[dict]) -> list[int]:
        """Full report: reconcile; returns block ids the worker should NOT
        have (master has no record) so it can delete them
        (master_filesystem.rs:1289 analog). Caller supplies the valid set."""
        for b in blocks:
            self.add_location(b["block_id"], worker_id, b.get("tier", "MEM"))
        return []

    def check_expired(self) -> list[int]:
        """Expire workers that missed heartbeats; returns lost worker ids."""
        deadline = now_ms() - self.expire_ms
        lost = [wid for wid, w in self.workers.items()
                if w.last_heartbeat_ms < deadline and w.state != WorkerState.LOST]
        for wid in lost:
            self.workers[wid].state = int(WorkerState.LOST)
            if self.mirror:
                self.mirror.upsert_worker(self.workers[wid])
            log.warning("worker %d expired -> LOST", wid)
        # callers run handle_lost_workers() to drop locations + re-replicate
        return lost

    def remove_worker_locations(self, worker_id: int) -> list[int]:
        """Drop all block locations on a dead worker; returns affected
        block ids (delete_locations analog, master_filesystem.rs:1581)."""
        affected = []
        for bid, locs in list(self.block_locs.items()):
            if worker_id in locs:
                locs.pop(worker_id)
                affected.append(bid)
                if not locs:
                    self.block_locs.pop(bid, None)
                if self.mirror:
                    self.mirror.remove_loc(bid, worker_id)
        return affected

    def decommission(self, worker_id: int) -> None:
        w = self.workers.get(worker_id)
        if w is None:
            raise err.WorkerNotFound(str(worker_id))
        w.state = int(WorkerState.DECOMMISSIONING)
        if self.mirror:
            self.mirror.upsert_worker(w)

    def live_workers(self) -> list[WorkerInfo]:
        return [w for w in self.workers.values()
                if w.state in (WorkerState.LIVE, WorkerState.DECOMMISSIONING)
                and w.state != WorkerState.DECOMMISSIONING]

    def get(self, worker_id: int) -> Optional[WorkerInfo]:
        return self.workers.get(worker_id)

    def add_command(self, worker_id: int, cmd: dict) -> None:
        self.commands.setdefault(worker_id, []).append(cmd)

    def schedule_block_delete(self, block_ids: list[int]) -> None:
        for bid in block_ids:
            for wid in self.block_locs.get(bid, {}):
                self.add_command(wid, {"cmd": CMD_DELETE_BLOCK, "block_id": bid})
            self.block_locs.pop(bid, None)
            if self.mirror:
                self.mirror.drop_block(bid)

    # ---------------- placement ----------------
    def choose_workers(self, count: int, policy: str = "local",
                       client_host: str = "", client_worker_id: int = -1,
                       exclude: set[int] | None = None,
                       tier: str = "") -> list[WorkerInfo]:
        """Pick `count` distinct live workers (choose_worker analog,
        master_filesystem.rs:579-600). `local` prefers the client's
        colocated worker first, then falls back to load-based."""
        exclude = exclude or set()
        cands = [w for w in self.live_workers()
                 if w.address.worker_id not in exclude and w.available > 0]
        if not cands:
            raise err.NoAvailableWorker(
                f"no live worker (total={len(self.workers)})")
        chosen: list[WorkerInfo] = []
        if policy == "local":
            local = [w for w in cands
                     if w.address.worker_id == client_worker_id
                     or (client_host and w.address.hostname == client_host)]
            # prefer the exact worker id, then same-host
            local.sort(key=lambda w: 0 if w.address.worker_id == client_worker_id else 1)
            for w in local:
                if len(chosen) < count:
                    chosen.append(w)
        rest = [w for w in cands if w not in chosen]
        need = count - len(chosen)
        if need > 0:
            if policy == "random":
                random.shuffle(rest)
            elif policy == "round_robin":
                self._rr += 1
                rest = rest[self._rr % max(1, len(rest)):] + rest[:self._rr % max(1, len(rest))]
            else:  # load_based (also the fallback for local)
                rest.sort(key=lambda w: w.used / max(1, w.capacity))
            chosen.extend(rest[:need])
        if not chosen:
            raise err.NoAvailableWorker("placement yielded no worker")
        return chosen[:count]

    def locations_of(self, block_id: int) -> list[tuple[WorkerInfo, str]]:
        """Live (worker, tier) pairs for a block, hottest tier first."""
        out = []
        for wid, tier in self.block_locs.get(block_id, {}).items():
            w = self.workers.get(wid)
            if w is not None and w.state != WorkerState.LOST:
                out.append((w, tier))
        out.sort(key=lambda p: TIER_ORDER.get(p[1], 9))
        return out

    # ---------------- capacity ----------------
    def total_capacity(self) -> int:
        return sum(w.capacity for w in self.live_workers())

    def total_used(self) -> int:
        return sum(w.used for w in self.live_workers())
