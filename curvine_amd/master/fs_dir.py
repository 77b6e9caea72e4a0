"""In-memory inode tree with journal emission on every mutation.

Analog of the reference's `FsDir`
(/root/reference/curvine-master/src/master/meta/fs_dir.rs:42-46; mkdir
:130-164, rename :346-368, acquire_new_block :703-731, complete_file
:737-763) and the `InodeView` file/dir variants (meta/inode/).

Every mutation is implemented as a pure ``_apply_*`` function driven by a
journal entry; public methods validate, build the entry, apply it and hand
it to the journal writer — so replay (`apply_entry`) and live execution are
the same deterministic code path (op_id ordering, fs_dir.rs:116-120).
"""
from __future__ import annotations

import logging
from typing import Iterator, Optional

from curvine_amd import errors as err
from curvine_amd.conf import TIER_MEM
from curvine_amd.master.journal import JournalWriter, Op
from curvine_amd.model import (BlockInfo, FileStatus, FileType, now_ms)

log = logging.getLogger("curvine.fsdir")

ROOT_ID = 1


class Inode:
    __slots__ = ("id", "parent_id", "name", "file_type", "is_dir", "children",
                 "length", "blocks", "block_size", "replicas", "storage_tier",
                 "complete", "mtime_ms", "atime_ms", "mode", "uid", "gid",
                 "ttl_ms", "ttl_action", "symlink_target", "nlink", "xattrs",
                 "create_ms", "access_count")

    def __init__(self, id: int, name: str, file_type: int, mode: int = 0o755):
        self.id = id
        self.parent_id = 0
        self.name = name
        self.file_type = file_type
        # plain attribute, not a property: checked several times per
        # mutation and file_type never changes after construction
        self.is_dir = file_type == FileType.DIR
        self.children: Optional[dict[str, int]] = {} if self.is_dir else None
        self.length = 0
        self.blocks: list[list[int]] = []   # [block_id, length]
        self.block_size = 64 << 20
        self.replicas = 1
        self.storage_tier = TIER_MEM
        self.complete = True
        t = now_ms()
        self.create_ms = t
        self.mtime_ms = t
        self.atime_ms = t
        self.mode = mode
        self.uid = 0
        self.gid = 0
        self.ttl_ms = 0
        self.ttl_action = "none"
        self.symlink_target = ""
        self.nlink = 1
        self.xattrs: dict[str, bytes] = {}
        self.access_count = 0

    def to_state(self) -> dict:
        return {s: (getattr(self, s) if s != "children" else
                    (dict(self.children) if self.children is not None else None))
                for s in self.__slots__}

    @staticmethod
    def from_state(d: dict) -> "Inode":
        ino = Inode(d["id"], d["name"], d["file_type"])
        for s in Inode.__slots__:
            if s in d:
                setattr(ino, s, d[s])
        ino.is_dir = ino.file_type == FileType.DIR
        if ino.is_dir and ino.children is None:
            ino.children = {}
        return ino


def norm_path(path: str) -> str:
    # fast path: already normalized (no "//", no "." or ".." segments —
    # any such segment necessarily contains "/." — no trailing slash).
    # Called several times per mutation; the rebuild below is the
    # exception, not the rule.
    if path.startswith("/") and "//" not in path and "/." not in path \
            and (len(path) == 1 or path[-1] != "/"):
        return path
    if not path.startswith("/"):
        raise err.InvalidPath(f"path must be absolute: {path!r}")
    parts = [p for p in path.split("/") if p and p != "."]
    for p in parts:
        if p == "..":
            raise err.InvalidPath(f"'..' not allowed: {path!r}")
    return "/" + "/".join(parts)


def split_path(path: str) -> tuple[str, str]:
    path = norm_path(path)
    if path == "/":
        return "/", ""
    parent, _, name = path.rpartition("/")
    return parent or "/", name


class MirrorFanout:
    """Several FsDir observers behind the single ``mirror`` slot (e.g.
    the native metadata mirror + the sqlite inode store)."""

    def __init__(self, mirrors: list):
        self.mirrors = mirrors

    def upsert(self, node) -> None:
        for m in self.mirrors:
            m.upsert(node)

    def touch(self, inode_id: int, mtime_ms: int) -> None:
        for m in self.mirrors:
            m.touch(inode_id, mtime_ms)

    def add_child(self, parent_id: int, name: str, child_id: int) -> None:
        for m in self.mirrors:
            m.add_child(parent_id, name, child_id)

    def remove_child(self, parent_id: int, name: str) -> None:
        for m in self.mirrors:
            m.remove_child(parent_id, name)

    def drop(self, inode_id: int) -> None:
        for m in self.mirrors:
            m.drop(inode_id)


class FsDir:
    def __init__(self, journal: JournalWriter):
        self.journal = journal
        self.inodes: dict[int, Inode] = {}
        self.block_index: dict[int, int] = {}   # block_id -> inode_id
        self.next_inode_id = ROOT_ID
        self.next_block_id = 0
        root = Inode(ROOT_ID, "", FileType.DIR, 0o755)
        self.inodes[ROOT_ID] = root
        # native metadata mirror (master/native_meta.MetaMirror): notified
        # synchronously from every _apply_* so the C++ read path stays
        # coherent with this tree
        self.mirror = None

    # ---------------- lookup ----------------
    def resolve(self, path: str) -> Optional[Inode]:
        path = norm_path(path)
        node = self.inodes[ROOT_ID]
        if path == "/":
            return node
        for part in path.strip("/").split("/"):
            if not node.is_dir or node.children is None:
                return None
            cid = node.children.get(part)
            if cid is None:
                return None
            node = self.inodes[cid]
        return node

    def must_resolve(self, path: str) -> Inode:
        node = self.resolve(path)
        if node is None:
            raise err.FileNotFound(path)
        return node

    def path_of(self, inode_id: int) -> str:
        parts: list[str] = []
        node = self.inodes.get(inode_id)
        while node is not None and node.id != ROOT_ID:
            parts.append(node.name)
            node = self.inodes.get(node.parent_id)
        return "/" + "/".join(reversed(parts))

    def status_of(self, node: Inode, path: str | None = None) -> FileStatus:
        return FileStatus(
            inode_id=node.id,
            path=path if path is not None else self.path_of(node.id),
            name=node.name, file_type=int(node.file_type),
            length=node.length, is_complete=node.complete,
            block_size=node.block_size, replicas=node.replicas,
            storage_tier=node.storage_tier,
            mtime_ms=node.mtime_ms, atime_ms=node.atime_ms,
            mode=node.mode, uid=node.uid, gid=node.gid,
            ttl_ms=node.ttl_ms, ttl_action=node.ttl_action,
            symlink_target=node.symlink_target, nlink=node.nlink,
            xattrs={k: bytes(v) for k, v in node.xattrs.items()})

    def status_dict(self, node: Inode, path: str | None = None) -> dict:
        """Reply-shaped status dict without the FileStatus dataclass hop
        (mutation-QPS hot path; keys == FileStatus.to_dict)."""
        return {
            "inode_id": node.id,
            "path": path if path is not None else self.path_of(node.id),
            "name": node.name, "file_type": int(node.file_type),
            "length": node.length, "is_complete": node.complete,
            "block_size": node.block_size, "replicas": node.replicas,
            "storage_tier": node.storage_tier,
            "mtime_ms": node.mtime_ms, "atime_ms": node.atime_ms,
            "mode": node.mode, "uid": node.uid, "gid": node.gid,
            "ttl_ms": node.ttl_ms, "ttl_action": node.ttl_action,
            "symlink_target": node.symlink_target, "nlink": node.nlink,
            "xattrs": node.xattrs,
        }

    def iter_files(self) -> Iterator[Inode]:
        for node in self.inodes.values():
            if node.file_type == FileType.FILE:
                yield node

    @staticmethod
    def _stamp(node: "Inode", e: dict) -> None:
        """Creation/mtime from the entry's log-time stamp: replay (WAL,
        raft followers, partial-flush restarts) reproduces identical
        timestamps instead of re-reading the clock."""
        ts = e.get("ts")
        if ts:
            node.create_ms = node.mtime_ms = node.atime_ms = ts

    # ---------------- mutations ----------------
    def mkdir(self, path: str, mode: int = 0o755, create_parents: bool = False) -> Inode:
        path = norm_path(path)
        node = self.resolve(path)
        if node is not None:
            if node.is_dir:
                return node
            raise err.FileAlreadyExists(path)
        parent_path, name = split_path(path)
        parent = self.resolve(parent_path)
        if parent is None:
            if not create_parents:
                raise err.FileNotFound(parent_path)
            parent = self.mkdir(parent_path, mode, True)
        if not parent.is_dir:
            raise err.NotDirectory(parent_path)
        entry = self.journal.log(Op.MKDIR, parent_id=parent.id, name=name,
                                 inode_id=self.next_inode_id + 1, mode=mode)
        return self._apply_mkdir(entry)

    def _apply_mkdir(self, e: dict) -> Inode:
        node = Inode(e["inode_id"], e["name"], FileType.DIR, e.get("mode", 0o755))
        self._stamp(node, e)
        node.parent_id = e["parent_id"]
        self.inodes[node.id] = node
        self.inodes[e["parent_id"]].children[e["name"]] = node.id
        self.next_inode_id = max(self.next_inode_id, node.id)
        if self.mirror:
            self.mirror.upsert(node)
            self.mirror.add_child(e["parent_id"], e["name"], node.id)
        return node

    def create(self, path: str, block_size: int, replicas: int,
               storage_tier: str, overwrite: bool = False,
               mode: int = 0o644, create_parents: bool = True,
               file_type: int = int(FileType.FILE)) -> tuple[Inode, list[int]]:
        """Returns (inode, blocks_to_delete_of_overwritten_file)."""
        path = norm_path(path)
        parent_path, name = split_path(path)
        parent = self.resolve(parent_path)
        if parent is None:
            if not create_parents:
                raise err.FileNotFound(parent_path)
            parent = self.mkdir(parent_path, 0o755, True)
        if not parent.is_dir:
            raise err.NotDirectory(parent_path)
        removed_blocks: list[int] = []
        child_id = parent.children.get(name)
        existing = self.inodes.get(child_id) if child_id is not None else None
        if existing is not None:
            if existing.is_dir:
                raise err.IsDirectory(path)
            if not overwrite:
                raise err.FileAlreadyExists(path)
        entry = self.journal.log(
            Op.CREATE, parent_id=parent.id, name=name,
            inode_id=self.next_inode_id + 1, block_size=block_size,
            replicas=replicas, storage_tier=storage_tier, mode=mode,
            overwrite=overwrite, file_type=file_type)
        node, removed_blocks = self._apply_create(entry)
        return node, removed_blocks

    def _apply_create(self, e: dict) -> tuple[Inode, list[int]]:
        parent = self.inodes[e["parent_id"]]
        removed: list[int] = []
        old_id = parent.children.get(e["name"])
        if old_id is not None and old_id != e["inode_id"] \
                and old_id in self.inodes:
            # overwrite of a different pre-existing file.  old_id ==
            # inode_id means idempotent replay of this very create; a
            # dangling edge (old_id not in inodes) can come from a
            # partially flushed store whose WAL tail we are replaying.
            removed = self._drop_inode(self.inodes[old_id])
        node = Inode(e["inode_id"], e["name"], e.get("file_type", int(FileType.FILE)),
                     e.get("mode", 0o644))
        self._stamp(node, e)
        node.parent_id = parent.id
        node.block_size = e["block_size"]
        node.replicas = e["replicas"]
        node.storage_tier = e["storage_tier"]
        node.complete = False
        self.inodes[node.id] = node
        parent.children[e["name"]] = node.id
        parent.mtime_ms = e.get("ts") or now_ms()
        self.next_inode_id = max(self.next_inode_id, node.id)
        if self.mirror:
            self.mirror.upsert(node)
            self.mirror.touch(parent.id, parent.mtime_ms)
            self.mirror.add_child(parent.id, e["name"], node.id)
        return node, removed

    def add_block(self, node: Inode, commit_prev_len: int = -1) -> int:
        """Allocate a new block id for an incomplete file
        (acquire_new_block analog). commit_prev_len >= 0 finalizes the
        previous block's length."""
        if node.complete:
            raise err.FileInWriting(f"{self.path_of(node.id)} is complete")
        entry = self.journal.log(Op.ADD_BLOCK, inode_id=node.id,
                                 block_id=self.next_block_id + 1,
                                 commit_prev_len=commit_prev_len)
        return self._apply_add_block(entry)

    def _apply_add_block(self, e: dict) -> int:
        node = self.inodes[e["inode_id"]]
        bid = e["block_id"]
        if any(b[0] == bid for b in node.blocks):
            # idempotent replay: WAL tail re-applied over a partially
            # flushed store already contains this block
            self.block_index[bid] = node.id
            self.next_block_id = max(self.next_block_id, bid)
            return bid
        if e.get("commit_prev_len", -1) >= 0 and node.blocks:
            node.blocks[-1][1] = e["commit_prev_len"]
        node.blocks.append([bid, 0])
        self.block_index[bid] = node.id
        self.next_block_id = max(self.next_block_id, bid)
        if self.mirror:
            self.mirror.upsert(node)   # block list feeds native open()
        return bid

    def complete_file(self, node: Inode, length: int,
                      block_lens: list[int] | None = None) -> None:
        entry = self.journal.log(Op.COMPLETE_FILE, inode_id=node.id,
                                 length=length, block_lens=block_lens)
        self._apply_complete(entry)

    def _apply_complete(self, e: dict) -> None:
        node = self.inodes[e["inode_id"]]
        node.length = e["length"]
        if e.get("block_lens"):
            for blk, ln in zip(node.blocks, e["block_lens"]):
                blk[1] = ln
        else:
            # derive block lengths from total length/block_size
            rem = node.length
            for blk in node.blocks:
                blk[1] = min(rem, node.block_size)
                rem -= blk[1]
        node.complete = True
        node.mtime_ms = e.get("ts") or now_ms()
        if self.mirror:
            self.mirror.upsert(node)

    def delete(self, path: str, recursive: bool = False) -> list[int]:
        """Returns deleted block ids (caller schedules worker deletes)."""
        path = norm_path(path)
        if path == "/":
            raise err.InvalidPath("cannot delete /")
        node = self.must_resolve(path)
        if node.is_dir and node.children and not recursive:
            raise err.DirNotEmpty(path)
        parent_path, name = split_path(path)
        parent = self.must_resolve(parent_path)
        entry = self.journal.log(Op.DELETE, inode_id=node.id,
                                 parent_id=parent.id, name=name)
        return self._apply_delete(entry)

    def _apply_delete(self, e: dict) -> list[int]:
        node = self.inodes.get(e["inode_id"])
        if node is None:
            return []
        # the journal entry records WHICH dentry was unlinked (hardlinks)
        parent = self.inodes.get(e.get("parent_id", node.parent_id))
        name = e.get("name", node.name)
        if parent is not None and parent.children is not None:
            parent.children.pop(name, None)
            parent.mtime_ms = e.get("ts") or now_ms()
            if self.mirror:
                self.mirror.remove_child(parent.id, name)
                self.mirror.touch(parent.id, parent.mtime_ms)
        if node.file_type == FileType.FILE and node.nlink > 1:
            node.nlink -= 1   # other hardlinked names keep the data
            if self.mirror:
                self.mirror.upsert(node)
            return []
        return self._drop_inode(node)

    def _drop_inode(self, node: Inode) -> list[int]:
        removed: list[int] = []
        stack = [node]
        while stack:
            n = stack.pop()
            if n.is_dir and n.children:
                stack.extend(self.inodes[c] for c in n.children.values())
            for bid, _ in n.blocks:
                self.block_index.pop(bid, None)
                removed.append(bid)
            self.inodes.pop(n.id, None)
            if self.mirror:
                self.mirror.drop(n.id)
        return removed

    def rename(self, src: str, dst: str) -> None:
        src, dst = norm_path(src), norm_path(dst)
        if src == "/" or dst == "/":
            raise err.InvalidPath("cannot rename /")
        if dst == src:
            return
        if dst.startswith(src + "/"):
            raise err.InvalidPath(f"cannot rename {src} into itself")
        node = self.must_resolve(src)
        src_parent_path, src_name = split_path(src)
        src_parent = self.must_resolve(src_parent_path)
        dst_parent_path, dst_name = split_path(dst)
        dst_parent = self.must_resolve(dst_parent_path)
        if not dst_parent.is_dir:
            raise err.NotDirectory(dst_parent_path)
        existing = self.resolve(dst)
        if existing is not None:
            if existing.is_dir:
                if existing.children:
                    raise err.DirNotEmpty(dst)
            elif node.is_dir:
                raise err.NotDirectory(dst)
        # record WHICH dentry moves: renaming a hardlink via a secondary
        # name must not touch the primary dentry (same rule as delete)
        entry = self.journal.log(Op.RENAME, inode_id=node.id,
                                 src_parent=src_parent.id, src_name=src_name,
                                 dst_parent=dst_parent.id, dst_name=dst_name)
        self._apply_rename(entry)

    def _apply_rename(self, e: dict) -> list[int]:
        node = self.inodes[e["inode_id"]]
        dst_parent = self.inodes[e["dst_parent"]]
        removed: list[int] = []
        old_id = dst_parent.children.get(e["dst_name"])
        if old_id is not None and old_id != node.id:
            removed = self._drop_inode(self.inodes[old_id])
        src_parent = self.inodes.get(e.get("src_parent", node.parent_id))
        old_name = e.get("src_name", node.name)
        if src_parent is not None and src_parent.children is not None:
            src_parent.children.pop(old_name, None)
            src_parent.mtime_ms = e.get("ts") or now_ms()
        node.parent_id = dst_parent.id
        node.name = e["dst_name"]
        dst_parent.children[node.name] = node.id
        dst_parent.mtime_ms = e.get("ts") or now_ms()
        if self.mirror:
            if src_parent is not None:
                self.mirror.remove_child(src_parent.id, old_name)
                self.mirror.touch(src_parent.id, src_parent.mtime_ms)
            self.mirror.upsert(node)            # name changed
            self.mirror.add_child(dst_parent.id, node.name, node.id)
            self.mirror.touch(dst_parent.id, dst_parent.mtime_ms)
        return removed

    def set_attr(self, node: Inode, **attrs) -> None:
        entry = self.journal.log(Op.SET_ATTR, inode_id=node.id, attrs=attrs)
        self._apply_set_attr(entry)

    def _apply_set_attr(self, e: dict) -> None:
        node = self.inodes.get(e["inode_id"])
        if node is None:
            return
        for k, v in e["attrs"].items():
            if k in ("mode", "uid", "gid", "atime_ms", "mtime_ms", "ttl_ms",
                     "ttl_action", "replicas", "storage_tier"):
                setattr(node, k, v)
        if self.mirror:
            self.mirror.upsert(node)

    def set_xattr(self, node: Inode, name: str, value: bytes) -> None:
        entry = self.journal.log(Op.SET_XATTR, inode_id=node.id,
                                 name=name, value=value)
        self._apply_set_xattr(entry)

    def _apply_set_xattr(self, e: dict) -> None:
        node = self.inodes.get(e["inode_id"])
        if node is not None:
            node.xattrs[e["name"]] = e["value"]
            if self.mirror:
                self.mirror.upsert(node)

    def remove_xattr(self, node: Inode, name: str) -> None:
        entry = self.journal.log(Op.REMOVE_XATTR, inode_id=node.id, name=name)
        self._apply_remove_xattr(entry)

    def _apply_remove_xattr(self, e: dict) -> None:
        node = self.inodes.get(e["inode_id"])
        if node is not None:
            node.xattrs.pop(e["name"], None)
            if self.mirror:
                self.mirror.upsert(node)

    def symlink(self, link_path: str, target: str) -> Inode:
        link_path = norm_path(link_path)
        if self.resolve(link_path) is not None:
            raise err.FileAlreadyExists(link_path)
        parent_path, name = split_path(link_path)
        parent = self.must_resolve(parent_path)
        if not parent.is_dir:
            raise err.NotDirectory(parent_path)
        entry = self.journal.log(Op.SYMLINK, parent_id=parent.id, name=name,
                                 inode_id=self.next_inode_id + 1, target=target)
        return self._apply_symlink(entry)

    def _apply_symlink(self, e: dict) -> Inode:
        node = Inode(e["inode_id"], e["name"], int(FileType.SYMLINK), 0o777)
        self._stamp(node, e)
        node.parent_id = e["parent_id"]
        node.symlink_target = e["target"]
        node.complete = True
        self.inodes[node.id] = node
        self.inodes[e["parent_id"]].children[e["name"]] = node.id
        self.next_inode_id = max(self.next_inode_id, node.id)
        if self.mirror:
            self.mirror.upsert(node)
            self.mirror.add_child(e["parent_id"], e["name"], node.id)
        return node

    def link(self, src: str, dst: str) -> Inode:
        """Hard link: a second dentry to the same inode."""
        src_node = self.must_resolve(src)
        if src_node.is_dir:
            raise err.IsDirectory(src)
        dst = norm_path(dst)
        if self.resolve(dst) is not None:
            raise err.FileAlreadyExists(dst)
        parent_path, name = split_path(dst)
        parent = self.must_resolve(parent_path)
        entry = self.journal.log(Op.LINK, inode_id=src_node.id,
                                 dst_parent=parent.id, dst_name=name)
        self._apply_link(entry)
        return src_node

    def _apply_link(self, e: dict) -> None:
        node = self.inodes[e["inode_id"]]
        parent = self.inodes[e["dst_parent"]]
        if parent.children.get(e["dst_name"]) == node.id:
            return   # idempotent replay: dentry already present
        parent.children[e["dst_name"]] = node.id
        node.nlink += 1
        parent.mtime_ms = e.get("ts") or now_ms()
        if self.mirror:
            self.mirror.upsert(node)
            self.mirror.touch(parent.id, parent.mtime_ms)
            self.mirror.add_child(parent.id, e["dst_name"], node.id)

    def resize(self, node: Inode, new_length: int) -> list[int]:
        """Truncate. Returns block ids fully beyond the new length."""
        entry = self.journal.log(Op.RESIZE, inode_id=node.id, length=new_length)
        return self._apply_resize(entry)

    def _apply_resize(self, e: dict) -> list[int]:
        node = self.inodes.get(e["inode_id"])
        if node is None:
            return []
        new_len = e["length"]
        removed: list[int] = []
        node.length = new_len
        keep = []
        off = 0
        for bid, blen in node.blocks:
            if off >= new_len:
                removed.append(bid)
                self.block_index.pop(bid, None)
            else:
                keep.append([bid, min(blen if blen else node.block_size,
                                      new_len - off)])
            off += blen if blen else node.block_size
        node.blocks = keep
        node.mtime_ms = e.get("ts") or now_ms()
        if self.mirror:
            self.mirror.upsert(node)
        return removed

    def free(self, node: Inode) -> list[int]:
        """Drop cached blocks, keep metadata (cv free / ttl 'free')."""
        entry = self.journal.log(Op.FREE, inode_id=node.id)
        return self._apply_free(entry)

    def _apply_free(self, e: dict) -> list[int]:
        node = self.inodes.get(e["inode_id"])
        if node is None:
            return []
        removed = [bid for bid, _ in node.blocks]
        for bid in removed:
            self.block_index.pop(bid, None)
        node.blocks = []
        # not complete anymore in the cache sense; length metadata kept
        if self.mirror:
            self.mirror.upsert(node)
        return removed

    # ---------------- replay & snapshot ----------------
    APPLY = {
        Op.MKDIR: "_apply_mkdir", Op.CREATE: "_apply_create",
        Op.ADD_BLOCK: "_apply_add_block", Op.COMPLETE_FILE: "_apply_complete",
        Op.DELETE: "_apply_delete", Op.RENAME: "_apply_rename",
        Op.SET_ATTR: "_apply_set_attr", Op.SYMLINK: "_apply_symlink",
        Op.LINK: "_apply_link", Op.RESIZE: "_apply_resize",
        Op.FREE: "_apply_free", Op.SET_XATTR: "_apply_set_xattr",
        Op.REMOVE_XATTR: "_apply_remove_xattr",
    }

    def apply_entry(self, e: dict) -> None:
        fn = self.APPLY.get(e["op"])
        if fn is None:
            return
        getattr(self, fn)(e)
        self.journal.op_id = max(self.journal.op_id, e["op_id"])

    def to_snapshot(self) -> dict:
        return {
            "op_id": self.journal.op_id,
            "next_inode_id": self.next_inode_id,
            "next_block_id": self.next_block_id,
            "inodes": [n.to_state() for n in self.inodes.values()],
        }

    def load_snapshot(self, state: dict) -> int:
        self.inodes = {}
        self.block_index = {}
        for s in state["inodes"]:
            node = Inode.from_state(s)
            self.inodes[node.id] = node
            for bid, _ in node.blocks:
                self.block_index[bid] = node.id
        self.next_inode_id = state["next_inode_id"]
        self.next_block_id = state["next_block_id"]
        self.journal.op_id = state["op_id"]
        return state["op_id"]
