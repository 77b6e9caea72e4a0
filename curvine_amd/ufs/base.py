"""UFS trait (curvine-ufs-api `UnderFs`/`AsyncChunkReader` analog)."""
from __future__ import annotations

from typing import Optional


class UfsReader:
    def read(self, size: int) -> bytes:
        raise NotImplementedError

    def seek(self, offset: int) -> None:
        raise NotImplementedError

    def close(self) -> None:
        pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class UfsWriter:
    def write(self, data: bytes) -> int:
        raise NotImplementedError

    def close(self) -> None:
        pass

    def __enter__(self):
        return self

    def __exit__(self, *a):
        self.close()


class UnderFs:
    """Paths are relative to the UFS root URI ('/a/b')."""

    scheme = "?"

    def list_files(self, path: str = "/", recursive: bool = True) -> list[dict]:
        """[{path, length, is_dir=False}...] files only."""
        raise NotImplementedError

    def status(self, path: str) -> Optional[dict]:
        raise NotImplementedError

    def open(self, path: str, offset: int = 0) -> UfsReader:
        raise NotImplementedError

    def create(self, path: str) -> UfsWriter:
        raise NotImplementedError

    def delete(self, path: str, recursive: bool = False) -> None:
        raise NotImplementedError

    def mkdir(self, path: str) -> None:
        raise NotImplementedError

    def rename(self, src: str, dst: str) -> None:
        raise NotImplementedError
