"""Journal: segmented write-ahead log + snapshots.

Analog of the reference's journal system
(/root/reference/curvine-master/src/master/journal/: `JournalEntry` 18
variants entry.rs:196-216, `JournalWriter.log_*` journal_writer.rs:35-264,
`JournalLoader.apply_entry` journal_loader.rs:616, snapshot create/apply
journal_system.rs:391-417).

Entries are msgpack maps with a monotonically increasing ``op_id`` (ordering
contract of fs_dir.rs:116-120), framed as::

    len: u32 BE | crc32c(payload): u32 BE | payload (msgpack)

Segments roll at ``segment_max_bytes``; a snapshot file captures the whole
FsDir state and allows truncating older segments.  When a raft group is
configured (journal.peers > 1), entries are proposed through
`curvine_amd.master.raft` instead of written locally; the same encode/apply
path is reused as the raft state machine.
"""
from __future__ import annotations

import io
import logging
import os
import struct
import zlib
from typing import Callable, Iterator, Optional

import msgpack

from curvine_amd.model import now_ms

log = logging.getLogger("curvine.journal")

_FRAME = struct.Struct(">II")


# entry kinds (JournalEntry variants analog)
class Op:
    MKDIR = "mkdir"
    CREATE = "create"
    ADD_BLOCK = "add_block"
    COMPLETE_FILE = "complete_file"
    DELETE = "delete"
    RENAME = "rename"
    SET_ATTR = "set_attr"
    SYMLINK = "symlink"
    LINK = "link"
    RESIZE = "resize"
    MOUNT = "mount"
    UNMOUNT = "unmount"
    UPDATE_MOUNT = "update_mount"
    FREE = "free"
    TTL_EXPIRE = "ttl_expire"
    NEXT_IDS = "next_ids"
    SET_XATTR = "set_xattr"
    REMOVE_XATTR = "remove_xattr"


def crc32c_sw(data: bytes) -> int:
    # host-side framing checksum; zlib crc32 (the GPU pipeline uses real
    # CRC32C via the native module — this is only for WAL integrity)
    return zlib.crc32(data) & 0xFFFFFFFF


def encode_entry(entry: dict) -> bytes:
    payload = msgpack.packb(entry, use_bin_type=True)
    return _FRAME.pack(len(payload), crc32c_sw(payload)) + payload


def decode_stream(f: io.BufferedReader) -> Iterator[dict]:
    while True:
        hdr = f.read(8)
        if len(hdr) < 8:
            return
        ln, crc = _FRAME.unpack(hdr)
        payload = f.read(ln)
        if len(payload) < ln or crc32c_sw(payload) != crc:
            log.warning("journal: truncated/corrupt tail entry, stopping replay")
            return
        yield msgpack.unpackb(payload, raw=False)


class JournalWriter:
    """Appends entries to the active segment. `propose` may be overridden
    (raft) — default is local durable append."""

    def __init__(self, journal_dir: str, segment_max_bytes: int = 256 << 20,
                 sync: bool = False):
        self.dir = journal_dir
        self.segment_max = segment_max_bytes
        self.sync = sync
        self.op_id = 0
        os.makedirs(journal_dir, exist_ok=True)
        self._f: Optional[io.BufferedWriter] = None
        self._seg_start_op = 0
        self._seg_bytes = 0
        self.enabled = True
        self.on_log = None   # optional hook(entry) after each append

    # ---- segment files: seg_<first_op_id>.wal ----
    def _segments(self) -> list[tuple[int, str]]:
        out = []
        for name in os.listdir(self.dir):
            if name.startswith("seg_") and name.endswith(".wal"):
                out.append((int(name[4:-4]), os.path.join(self.dir, name)))
        return sorted(out)

    def _roll(self, first_op: int) -> None:
        if self._f:
            self._f.flush()
            os.fsync(self._f.fileno())
            self._f.close()
        path = os.path.join(self.dir, f"seg_{first_op:020d}.wal")
        self._f = open(path, "ab")
        self._seg_start_op = first_op
        self._seg_bytes = self._f.tell()

    def log(self, op: str, **fields) -> dict:
        if not self.enabled:
            return {}
        self.op_id += 1
        entry = {"op": op, "op_id": self.op_id, "ts": now_ms(), **fields}
        buf = encode_entry(entry)
        if self._f is None or self._seg_bytes + len(buf) > self.segment_max:
            self._roll(self.op_id)
        self._f.write(buf)
        self._seg_bytes += len(buf)
        if self.sync:
            self._f.flush()
            os.fsync(self._f.fileno())
        if self.on_log is not None:
            self.on_log(entry)
        return entry

    def flush(self) -> None:
        if self._f:
            self._f.flush()
            os.fsync(self._f.fileno())

    def close(self) -> None:
        if self._f:
            self.flush()
            self._f.close()
            self._f = None

    def purge_through(self, op_id: int) -> None:
        """Delete segments entirely covered by a snapshot at op_id."""
        segs = self._segments()
        for i, (first, path) in enumerate(segs):
            nxt = segs[i + 1][0] if i + 1 < len(segs) else None
            if nxt is not None and nxt <= op_id + 1:
                os.remove(path)


class JournalLoader:
    """Replays snapshot + WAL segments into an apply callback."""

    def __init__(self, journal_dir: str):
        self.dir = journal_dir

    def snapshot_path(self) -> str:
        return os.path.join(self.dir, "snapshot.bin")

    def load(self, apply_entry: Callable[[dict], None],
             load_snapshot: Callable[[dict], int],
             start_op: int = 0) -> int:
        """Returns the last op_id applied.  ``start_op`` is pre-restored
        state (e.g. the sqlite inode store): the snapshot is only loaded
        when it is newer, and WAL replay begins past whichever won."""
        last_op = start_op
        snap = self.snapshot_path()
        if os.path.exists(snap):
            with open(snap, "rb") as f:
                entries = list(decode_stream(f))
            if entries and entries[0].get("op_id", 0) > start_op:
                last_op = load_snapshot(entries[0])
        if not os.path.isdir(self.dir):
            return last_op
        segs = sorted(
            (int(n[4:-4]), os.path.join(self.dir, n))
            for n in os.listdir(self.dir)
            if n.startswith("seg_") and n.endswith(".wal"))
        for _, path in segs:
            with open(path, "rb") as f:
                for entry in decode_stream(f):
                    if entry["op_id"] <= last_op:
                        continue
                    apply_entry(entry)
                    last_op = entry["op_id"]
        return last_op

    def save_snapshot(self, state: dict) -> None:
        snap = self.snapshot_path()
        tmp = snap + ".tmp"
        os.makedirs(self.dir, exist_ok=True)
        with open(tmp, "wb") as f:
            f.write(encode_entry(state))
            f.flush()
            os.fsync(f.fileno())
        os.replace(tmp, snap)
