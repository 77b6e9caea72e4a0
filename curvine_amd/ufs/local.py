"""Local-directory UFS (testing + node-local cold storage)."""
from __future__ import annotations

import os
import shutil
from typing import Optional

from curvine_amd import errors as err
from curvine_amd.ufs.base import UfsReader, UfsWriter, UnderFs


class _LocalReader(UfsReader):
    def __init__(self, path: str, offset: int):
        self.f = open(path, "rb")
        if offset:
            self.f.seek(offset)

    def read(self, size: int) -> bytes:
        return self.f.read(size)

    def seek(self, offset: int) -> None:
        self.f.seek(offset)

    def close(self) -> None:
        self.f.close()


class _LocalWriter(UfsWriter):
    def __init__(self, path: str):
        os.makedirs(os.path.dirname(path), exist_ok=True)
        self.path = path
        self.f = open(path + ".tmp", "wb")

    def write(self, data: bytes) -> int:
        return self.f.write(data)

    def close(self) -> None:
        self.f.close()
        os.replace(self.path + ".tmp", self.path)


class LocalUfs(UnderFs):
    scheme = "file"

    def __init__(self, root: str):
        self.root = root.rstrip("/") or "/"

    def _abs(self, path: str) -> str:
        path = "/" + path.strip("/")
        full = os.path.normpath(self.root + path)
        if not full.startswith(self.root):
            raise err.InvalidPath(path)
        return full

    def list_files(self, path: str = "/", recursive: bool = True) -> list[dict]:
        base = self._abs(path)
        out: list[dict] = []
        if os.path.isfile(base):
            rel = "/" + os.path.relpath(base, self.root)
            return [{"path": rel, "length": os.path.getsize(base), "is_dir": False}]
        if not os.path.isdir(base):
            raise err.FileNotFound(path)
        if recursive:
            for dirpath, _, files in os.walk(base):
                for name in sorted(files):
                    full = os.path.join(dirpath, name)
                    rel = "/" + os.path.relpath(full, self.root)
                    out.append({"path": rel, "length": os.path.getsize(full),
                                "is_dir": False})
        else:
            for name in sorted(os.listdir(base)):
                full = os.path.join(base, name)
                rel = "/" + os.path.relpath(full, self.root)
                out.append({"path": rel, "length":
                            os.path.getsize(full) if os.path.isfile(full) else 0,
                            "is_dir": os.path.isdir(full)})
        return [f for f in out if not f["is_dir"]] if recursive else out

    def status(self, path: str) -> Optional[dict]:
        full = self._abs(path)
        if not os.path.exists(full):
            return None
        st = os.stat(full)
        return {"path": "/" + path.strip("/"), "length": st.st_size,
                "is_dir": os.path.isdir(full), "mtime_ms": int(st.st_mtime * 1000)}

    def open(self, path: str, offset: int = 0) -> UfsReader:
        full = self._abs(path)
        if not os.path.isfile(full):
            raise err.FileNotFound(path)
        return _LocalReader(full, offset)

    def create(self, path: str) -> UfsWriter:
        return _LocalWriter(self._abs(path))

    def delete(self, path: str, recursive: bool = False) -> None:
        full = self._abs(path)
        if os.path.isdir(full):
            if recursive:
                shutil.rmtree(full)
            else:
                os.rmdir(full)
        elif os.path.exists(full):
            os.remove(full)

    def mkdir(self, path: str) -> None:
        os.makedirs(self._abs(path), exist_ok=True)

    def rename(self, src: str, dst: str) -> None:
        os.makedirs(os.path.dirname(self._abs(dst)), exist_ok=True)
        os.replace(self._abs(src), self._abs(dst))
