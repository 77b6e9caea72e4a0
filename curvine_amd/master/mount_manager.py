"""UFS mount table.

Analog of /root/reference/curvine-master/src/master/mount/mount_manager.rs
(:27-171): mount points map a curvine path prefix to an under-filesystem
URI; persisted through the journal (Mount/UnMount entries).
"""
from __future__ import annotations


from curvine_amd import errors as err
from curvine_amd.master.fs_dir import norm_path
from curvine_amd.master.journal import JournalWriter, Op
from curvine_amd.model import MountInfo


class MountManager:
    def __init__(self, journal: JournalWriter):
        self.journal = journal
        self.mounts: dict[str, MountInfo] = {}   # curvine_path -> info
        self.next_id = 0

    def mount(self, curvine_path: str, ufs_path: str, properties: dict,
              cache_mode: str = "cache", auto_cache: bool = True) -> MountInfo:
        curvine_path = norm_path(curvine_path)
        for p in self.mounts:
            if p == curvine_path or p.startswith(curvine_path + "/") \
                    or curvine_path.startswith(p + "/"):
                raise err.FileAlreadyExists(f"overlapping mount {p}")
        entry = self.journal.log(Op.MOUNT, mount_id=self.next_id + 1,
                                 curvine_path=curvine_path, ufs_path=ufs_path,
                                 properties=properties, cache_mode=cache_mode,
                                 auto_cache=auto_cache)
        return self.apply_mount(entry)

    def apply_mount(self, e: dict) -> MountInfo:
        mi = MountInfo(mount_id=e["mount_id"], curvine_path=e["curvine_path"],
                       ufs_path=e["ufs_path"], properties=e.get("properties", {}),
                       cache_mode=e.get("cache_mode", "cache"),
                       auto_cache=e.get("auto_cache", True))
        self.mounts[mi.curvine_path] = mi
        self.next_id = max(self.next_id, mi.mount_id)
        return mi

    def unmount(self, curvine_path: str) -> None:
        curvine_path = norm_path(curvine_path)
        if curvine_path not in self.mounts:
            raise err.MountNotFound(curvine_path)
        entry = self.journal.log(Op.UNMOUNT, curvine_path=curvine_path)
        self.apply_unmount(entry)

    def apply_unmount(self, e: dict) -> None:
        self.mounts.pop(e["curvine_path"], None)

    def update(self, curvine_path: str, properties: dict | None,
               cache_mode: str | None, auto_cache: bool | None) -> MountInfo:
        curvine_path = norm_path(curvine_path)
        mi = self.mounts.get(curvine_path)
        if mi is None:
            raise err.MountNotFound(curvine_path)
        entry = self.journal.log(Op.UPDATE_MOUNT, curvine_path=curvine_path,
                                 properties=properties, cache_mode=cache_mode,
                                 auto_cache=auto_cache)
        return self.apply_update(entry)

    def apply_update(self, e: dict) -> MountInfo:
        mi = self.mounts.get(e["curvine_path"])
        if mi is None:
            raise err.MountNotFound(e["curvine_path"])
        if e.get("properties"):
            mi.properties.update(e["properties"])
        if e.get("cache_mode") is not None:
            mi.cache_mode = e["cache_mode"]
        if e.get("auto_cache") is not None:
            mi.auto_cache = e["auto_cache"]
        return mi

    def lookup(self, path: str):
        """Longest-prefix mount for a curvine path."""
        path = norm_path(path)
        best = None
        for p, mi in self.mounts.items():
            if path == p or path.startswith(p + "/"):
                if best is None or len(p) > len(best.curvine_path):
                    best = mi
        return best

    def table(self) -> list[MountInfo]:
        return list(self.mounts.values())

    def to_snapshot(self) -> list[dict]:
        return [m.to_dict() for m in self.mounts.values()]

    def load_snapshot(self, rows: list[dict]) -> None:
        self.mounts = {}
        for r in rows:
            mi = MountInfo.from_dict(r)
            self.mounts[mi.curvine_path] = mi
            self.next_id = max(self.next_id, mi.mount_id)

    def apply_entry(self, e: dict) -> bool:
        op = e.get("op")
        if op == Op.MOUNT:
            self.apply_mount(e)
        elif op == Op.UNMOUNT:
            self.apply_unmount(e)
        elif op == Op.UPDATE_MOUNT:
            self.apply_update(e)
        else:
            return False
        return True
