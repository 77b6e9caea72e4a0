"""Sqlite-backed durable inode store.

The reference keeps its inode tree in RocksDB (the
`RocksInodeStore` behind FsDir), giving incremental durability and
journal-free cold restarts.  This is the same capability on sqlite (the
KV engine this image ships): a mirror observer marks inodes dirty as
``FsDir._apply_*`` runs, and ``flush()`` batches them into one WAL-mode
transaction — so restart = one table scan + WAL-tail replay, not a full
journal replay.

Scope: the working set stays in memory (FsDir's dict); the DB is the
durability/restart layer.  Raft masters rebuild from the raft log and
don't use this store.
"""
from __future__ import annotations

import logging
import os
import sqlite3
from typing import Optional

import msgpack

log = logging.getLogger("curvine.inodedb")

_SCHEMA = """
CREATE TABLE IF NOT EXISTS inodes (id INTEGER PRIMARY KEY, state BLOB);
CREATE TABLE IF NOT EXISTS meta (k TEXT PRIMARY KEY, v BLOB);
"""


class SqliteInodeStore:
    """FsDir mirror + durable store.  Mirror hooks only mark dirt; the
    actual rows are written by flush() on the actor tick / shutdown."""

    def __init__(self, path: str):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        self.path = path
        self.conn = sqlite3.connect(path, check_same_thread=False)
        self.conn.executescript(_SCHEMA)
        self.conn.execute("PRAGMA journal_mode=WAL")
        self.conn.execute("PRAGMA synchronous=NORMAL")
        self._dirty: set[int] = set()
        self._deleted: set[int] = set()

    # ---------------- mirror interface (same as MetaMirror) ----------------
    def upsert(self, node) -> None:
        self._dirty.add(node.id)
        self._deleted.discard(node.id)

    def touch(self, inode_id: int, mtime_ms: int) -> None:
        # mtime lives inside the serialized state: re-flush the row
        self._dirty.add(inode_id)

    def add_child(self, parent_id: int, name: str, child_id: int) -> None:
        # children live inside the parent's serialized state
        self._dirty.add(parent_id)

    def remove_child(self, parent_id: int, name: str) -> None:
        self._dirty.add(parent_id)

    def drop(self, inode_id: int) -> None:
        self._dirty.discard(inode_id)
        self._deleted.add(inode_id)

    # ---------------- persistence ----------------
    MAX_BATCH = 20_000   # bound the per-tick pause on mutation storms

    def flush(self, fs_dir, mounts_state, op_id: int) -> int:
        """One transaction: dirty upserts + deletes + watermarks, capped
        at MAX_BATCH rows (the remainder stays dirty for the next tick —
        correctness is carried by the WAL tail + restart reconcile).
        Returns the number of rows written."""
        if not self._dirty and not self._deleted:
            cur = self.conn.execute("SELECT v FROM meta WHERE k='op_id'")
            row = cur.fetchone()
            if row is not None and int.from_bytes(row[0], "little") == op_id:
                return 0
        rows = []
        batch = []
        for iid in self._dirty:
            batch.append(iid)
            node = fs_dir.inodes.get(iid)
            if node is None:
                continue
            rows.append((iid, msgpack.packb(node.to_state(),
                                            use_bin_type=True)))
            if len(rows) >= self.MAX_BATCH:
                break
        partial = len(batch) < len(self._dirty)
        dels = [(iid,) for iid in self._deleted]
        with self.conn:
            if rows:
                self.conn.executemany(
                    "REPLACE INTO inodes (id, state) VALUES (?, ?)", rows)
            if dels:
                self.conn.executemany("DELETE FROM inodes WHERE id=?", dels)
            meta = [("next_inode_id",
                     fs_dir.next_inode_id.to_bytes(8, "little")),
                    ("next_block_id",
                     fs_dir.next_block_id.to_bytes(8, "little")),
                    ("mounts", msgpack.packb(mounts_state,
                                             use_bin_type=True))]
            if not partial:
                # advance the restart watermark only once every dirty
                # inode as of this op is actually on disk
                meta.append(("op_id", op_id.to_bytes(8, "little")))
            self.conn.executemany(
                "REPLACE INTO meta (k, v) VALUES (?, ?)", meta)
        n = len(rows) + len(dels)
        self._dirty.difference_update(batch)
        self._deleted.clear()
        return n

    def load(self, fs_dir, mounts) -> Optional[int]:
        """Populate fs_dir (and the mount table) from the DB; returns the
        stored op_id, or None when the DB is empty."""
        from curvine_amd.master.fs_dir import Inode
        cur = self.conn.execute("SELECT v FROM meta WHERE k='op_id'")
        row = cur.fetchone()
        if row is None:
            return None
        op_id = int.from_bytes(row[0], "little")
        fs_dir.inodes = {}
        fs_dir.block_index = {}
        n = 0
        for iid, state in self.conn.execute("SELECT id, state FROM inodes"):
            node = Inode.from_state(msgpack.unpackb(state, raw=False))
            fs_dir.inodes[node.id] = node
            for bid, _ in node.blocks:
                fs_dir.block_index[bid] = node.id
            n += 1
        for k in ("next_inode_id", "next_block_id"):
            r = self.conn.execute("SELECT v FROM meta WHERE k=?",
                                  (k,)).fetchone()
            if r is not None:
                setattr(fs_dir, k, int.from_bytes(r[0], "little"))
        r = self.conn.execute("SELECT v FROM meta WHERE k='mounts'").fetchone()
        if r is not None and mounts is not None:
            mounts.load_snapshot(msgpack.unpackb(r[0], raw=False))
        fs_dir.journal.op_id = op_id
        log.info("inode db: loaded %d inodes at op %d", n, op_id)
        return op_id

    def resync(self, fs_dir) -> None:
        """Full reconcile after out-of-band state changes (WAL tail
        replayed without the mirror, first enable on an existing
        namespace): every live inode dirty, every stale row deleted."""
        live = set(fs_dir.inodes.keys())
        db_ids = {r[0] for r in self.conn.execute("SELECT id FROM inodes")}
        self._deleted.update(db_ids - live)
        self._dirty.update(live)

    def close(self) -> None:
        try:
            self.conn.close()
        except Exception:  # noqa: BLE001
            pass
