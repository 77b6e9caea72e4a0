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
CREATE TABLE IF NOT EXISTS inodes (id INTEGER PRIMARY KEY, state BLOB,
    atime INTEGER DEFAULT 0, ttl_deadline INTEGER DEFAULT 0,
    is_file INTEGER DEFAULT 0);
CREATE TABLE IF NOT EXISTS blocks (block_id INTEGER PRIMARY KEY,
    inode_id INTEGER);
CREATE INDEX IF NOT EXISTS blocks_by_inode ON blocks (inode_id);
CREATE TABLE IF NOT EXISTS meta (k TEXT PRIMARY KEY, v BLOB);
"""


class PagedInodeMap(dict):
    """Bounded resident inode map: misses fault the row in from sqlite
    (RocksInodeStore paging analog, rocks_inode_store.rs:26-215).  The
    actor tick evicts cold, clean, flushed inodes via
    SqliteInodeStore.page_out."""

    __slots__ = ("loader",)

    def __init__(self, loader):
        super().__init__()
        self.loader = loader

    def __missing__(self, iid):
        node = self.loader(iid)
        if node is None:
            raise KeyError(iid)
        self[iid] = node
        return node

    def get(self, iid, default=None):
        try:
            return self[iid]
        except KeyError:
            return default

    def __contains__(self, iid):
        if super().__contains__(iid):
            return True
        return self.get(iid) is not None


class PagedBlockIndex(dict):
    """block_id -> inode_id with sqlite fallback for paged-out files."""

    __slots__ = ("conn",)

    def __init__(self, conn):
        super().__init__()
        self.conn = conn

    def __missing__(self, bid):
        row = self.conn.execute(
            "SELECT inode_id FROM blocks WHERE block_id=?", (bid,)
        ).fetchone()
        if row is None:
            raise KeyError(bid)
        return row[0]   # not cached: resident entries track resident nodes

    def get(self, bid, default=None):
        try:
            return self[bid]
        except KeyError:
            return default

    def __contains__(self, bid):
        if super().__contains__(bid):
            return True
        return self.get(bid) is not None


class SqliteInodeStore:
    """FsDir mirror + durable store.  Mirror hooks only mark dirt; the
    actual rows are written by flush() on the actor tick / shutdown."""

    def __init__(self, path: str):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        self.path = path
        self.conn = sqlite3.connect(path, check_same_thread=False)
        self._migrate()
        self.conn.executescript(_SCHEMA)
        self.conn.execute("PRAGMA journal_mode=WAL")
        self.conn.execute("PRAGMA synchronous=NORMAL")
        self._dirty: set[int] = set()
        self._deleted: set[int] = set()
        # async-flush state: rows snapshot on the master thread, the
        # sqlite transaction runs on a dedicated writer thread with its
        # own connection (WAL: one writer + concurrent readers), so the
        # actor tick never stalls mutations behind a 20k-row commit
        self._inflight: set[int] = set()
        self._pool = None
        self._commit_f = None
        self._wconn = None

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

    def _migrate(self) -> None:
        try:
            cols = {r[1] for r in self.conn.execute(
                "PRAGMA table_info(inodes)")}
        except sqlite3.Error:
            return
        if cols and "atime" not in cols:
            for ddl in ("ALTER TABLE inodes ADD COLUMN atime INTEGER DEFAULT 0",
                        "ALTER TABLE inodes ADD COLUMN ttl_deadline INTEGER DEFAULT 0",
                        "ALTER TABLE inodes ADD COLUMN is_file INTEGER DEFAULT 0"):
                try:
                    self.conn.execute(ddl)
                except sqlite3.Error:
                    pass

    # ---------------- persistence ----------------
    MAX_BATCH = 20_000   # bound the per-tick pause on mutation storms

    def flush(self, fs_dir, mounts_state, op_id: int) -> int:
        """Synchronous flush: snapshot + one transaction on the calling
        thread (shutdown / checkpoint path).  Waits out any in-flight
        async commit first so row versions never go backwards."""
        self.wait_flush()
        snap = self._snapshot(fs_dir, mounts_state, op_id)
        if snap is None:
            return 0
        try:
            return self._commit(self.conn, snap)
        except Exception:
            self._recover(snap)
            raise
        finally:
            self._inflight.clear()

    def flush_async(self, fs_dir, mounts_state, op_id: int) -> bool:
        """Actor-tick flush: snapshot rows on the caller (master) thread —
        the consistent-read part — then commit them on the writer thread.
        Single-flight: a tick that lands while a commit is still running
        skips (the dirt just waits for the next tick)."""
        if self._commit_f is not None and not self._commit_f.done():
            return False
        snap = self._snapshot(fs_dir, mounts_state, op_id)
        if snap is None:
            return False
        if self._pool is None:
            from concurrent.futures import ThreadPoolExecutor
            self._pool = ThreadPoolExecutor(
                max_workers=1, thread_name_prefix="inode-db-flush")
        self._commit_f = self._pool.submit(self._commit_async, snap)
        return True

    def wait_flush(self) -> None:
        """Block until any in-flight async commit lands."""
        f, self._commit_f = self._commit_f, None
        if f is not None:
            try:
                f.result(timeout=60)
            except Exception:  # noqa: BLE001 — already recovered to dirty
                pass

    def _commit_async(self, snap) -> int:
        if self._wconn is None:
            self._wconn = sqlite3.connect(self.path, check_same_thread=False)
            self._wconn.execute("PRAGMA synchronous=NORMAL")
        try:
            return self._commit(self._wconn, snap)
        except Exception:  # noqa: BLE001
            log.exception("async inode-db flush failed; rows stay dirty")
            self._recover(snap)
            return 0
        finally:
            self._inflight.clear()

    def _recover(self, snap) -> None:
        rows, block_rows, dels, batch, partial, op_id, mounts_state = snap
        self._dirty.update(batch)
        self._deleted.update(d[0] for d in dels)

    def _snapshot(self, fs_dir, mounts_state, op_id: int):
        """Consistent snapshot of up to MAX_BATCH dirty rows; must run on
        the mutating (master) thread.  Clears the taken dirt — a failure
        path re-adds it — and marks it in flight so page_out will not
        evict a node whose row is not committed yet."""
        if not self._dirty and not self._deleted:
            cur = self.conn.execute("SELECT v FROM meta WHERE k='op_id'")
            row = cur.fetchone()
            if row is not None and int.from_bytes(row[0], "little") == op_id:
                return None
        rows = []
        batch = []
        block_rows = []
        resident = dict.get   # bypass paging: a dirty inode IS resident
        for iid in self._dirty:
            batch.append(iid)
            node = resident(fs_dir.inodes, iid)
            if node is None:
                continue
            ttl_deadline = (node.create_ms + node.ttl_ms) \
                if node.ttl_ms > 0 else 0
            rows.append((iid, msgpack.packb(node.to_state(),
                                            use_bin_type=True),
                         node.atime_ms, ttl_deadline,
                         0 if node.is_dir else 1))
            for bid, _ in node.blocks:
                block_rows.append((bid, iid))
            if len(rows) >= self.MAX_BATCH:
                break
        partial = len(batch) < len(self._dirty)
        dels = [(iid,) for iid in self._deleted]
        self._dirty.difference_update(batch)
        self._deleted.clear()
        self._inflight = set(batch) | {d[0] for d in dels}
        # watermarks are part of the snapshot (the writer thread must not
        # read fs_dir)
        mstate = (fs_dir.next_inode_id, fs_dir.next_block_id, mounts_state)
        return rows, block_rows, dels, batch, partial, op_id, mstate

    def _commit(self, conn, snap) -> int:
        rows, block_rows, dels, batch, partial, op_id, mstate = snap
        next_inode_id, next_block_id, mounts_state = mstate
        with conn:
            if rows:
                conn.executemany(
                    "REPLACE INTO inodes (id, state, atime, ttl_deadline,"
                    " is_file) VALUES (?, ?, ?, ?, ?)", rows)
                # block index rows: drop-then-insert per dirty inode so
                # truncated files lose their stale entries
                conn.executemany(
                    "DELETE FROM blocks WHERE inode_id=?",
                    [(r[0],) for r in rows])
                if block_rows:
                    conn.executemany(
                        "REPLACE INTO blocks (block_id, inode_id)"
                        " VALUES (?, ?)", block_rows)
            if dels:
                conn.executemany("DELETE FROM inodes WHERE id=?", dels)
                conn.executemany(
                    "DELETE FROM blocks WHERE inode_id=?", dels)
            meta = [("next_inode_id", next_inode_id.to_bytes(8, "little")),
                    ("next_block_id", next_block_id.to_bytes(8, "little")),
                    ("mounts", msgpack.packb(mounts_state,
                                             use_bin_type=True))]
            if not partial:
                # advance the restart watermark only once every dirty
                # inode as of this op is actually on disk
                meta.append(("op_id", op_id.to_bytes(8, "little")))
            conn.executemany(
                "REPLACE INTO meta (k, v) VALUES (?, ?)", meta)
        return len(rows) + len(dels)

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

    # ---------------- paging (beyond-RAM namespace) ----------------
    on_fault = None   # hook(node): re-prime the native meta mirror

    def load_one(self, iid: int):
        """Fault a single inode in from its row (PagedInodeMap loader)."""
        from curvine_amd.master.fs_dir import Inode
        row = self.conn.execute(
            "SELECT state FROM inodes WHERE id=?", (iid,)).fetchone()
        if row is None:
            return None
        node = Inode.from_state(msgpack.unpackb(row[0], raw=False))
        if self.on_fault is not None:
            self.on_fault(node)
        return node

    def load_paged(self, fs_dir, mounts) -> Optional[int]:
        """Paged restart: restore watermarks + the root inode only; the
        rest of the namespace faults in on resolve."""
        from curvine_amd.master.fs_dir import ROOT_ID
        cur = self.conn.execute("SELECT v FROM meta WHERE k='op_id'")
        row = cur.fetchone()
        if row is None:
            return None
        op_id = int.from_bytes(row[0], "little")
        fs_dir.inodes = PagedInodeMap(self.load_one)
        fs_dir.block_index = PagedBlockIndex(self.conn)
        root = self.load_one(ROOT_ID)
        if root is not None:
            fs_dir.inodes[ROOT_ID] = root
        for k in ("next_inode_id", "next_block_id"):
            r = self.conn.execute("SELECT v FROM meta WHERE k=?",
                                  (k,)).fetchone()
            if r is not None:
                setattr(fs_dir, k, int.from_bytes(r[0], "little"))
        r = self.conn.execute("SELECT v FROM meta WHERE k='mounts'").fetchone()
        if r is not None and mounts is not None:
            mounts.load_snapshot(msgpack.unpackb(r[0], raw=False))
        fs_dir.journal.op_id = op_id
        n = self.conn.execute("SELECT COUNT(*) FROM inodes").fetchone()[0]
        log.info("inode db (paged): %d inodes on disk at op %d", n, op_id)
        return op_id

    def enable_paging(self, fs_dir) -> None:
        """Swap the plain dict maps for paged ones (fresh-DB boot)."""
        if isinstance(fs_dir.inodes, PagedInodeMap):
            return
        pm = PagedInodeMap(self.load_one)
        pm.update(fs_dir.inodes)
        bi = PagedBlockIndex(self.conn)
        bi.update(fs_dir.block_index)
        fs_dir.inodes, fs_dir.block_index = pm, bi

    def page_out(self, fs_dir, protected: set, max_resident: int,
                 on_evict=None) -> int:
        """Evict cold, CLEAN (flushed), unprotected inodes from the
        resident map down to ~90%% of max_resident.  Returns evictions.
        ``on_evict(iid)`` lets the native meta mirror drop its copy (a
        native lookup of an evicted path falls back to the Python
        handler, which faults the row back in)."""
        from curvine_amd.master.fs_dir import ROOT_ID
        inodes = fs_dir.inodes
        if not isinstance(inodes, PagedInodeMap) or \
                len(inodes) <= max_resident:
            return 0
        target = max(1, int(max_resident * 0.9))
        cands = [n for iid, n in inodes.items()
                 if iid != ROOT_ID and iid not in self._dirty
                 and iid not in self._deleted and iid not in self._inflight
                 and iid not in protected]
        cands.sort(key=lambda n: n.atime_ms)
        evicted = 0
        bindex = fs_dir.block_index
        for node in cands:
            if len(inodes) <= target:
                break
            dict.pop(inodes, node.id, None)
            for bid, _ in node.blocks:
                dict.pop(bindex, bid, None)
            if on_evict is not None:
                on_evict(node.id)
            evicted += 1
        return evicted

    def ttl_expired_ids(self, now_ms: int, limit: int = 10_000) -> list:
        return [r[0] for r in self.conn.execute(
            "SELECT id FROM inodes WHERE ttl_deadline > 0 AND"
            " ttl_deadline < ? LIMIT ?", (now_ms, limit))]

    def cold_file_ids(self, limit: int = 10_000) -> list:
        """Files ordered by (flushed) atime — capacity-eviction input."""
        return [r[0] for r in self.conn.execute(
            "SELECT id FROM inodes WHERE is_file=1 ORDER BY atime"
            " LIMIT ?", (limit,))]

    def resync(self, fs_dir) -> None:
        """Full reconcile after out-of-band state changes (WAL tail
        replayed without the mirror, first enable on an existing
        namespace): every live inode dirty, every stale row deleted."""
        live = set(fs_dir.inodes.keys())
        db_ids = {r[0] for r in self.conn.execute("SELECT id FROM inodes")}
        self._deleted.update(db_ids - live)
        self._dirty.update(live)

    def close(self) -> None:
        self.wait_flush()
        if self._pool is not None:
            self._pool.shutdown(wait=True)
            self._pool = None
        for c in (self._wconn, self.conn):
            try:
                if c is not None:
                    c.close()
            except Exception:  # noqa: BLE001
                pass
        self._wconn = None
