"""PyTorch DataLoader integration: datasets over cached files.

The training-data path (BASELINE config[4]): WebDataset-style tar shards
cached in HBM/host tiers, consumed by a torch DataLoader.  Two entries:

* `CurvineShardDataset` — iterable dataset over tar shards stored in the
  cache (reads through the client short-circuit; optionally lands sample
  bytes straight into device tensors).
* `CurvineFileDataset` — map-style dataset of whole files.

Multiprocess loading: pass ``worker_init_fn=curvine_worker_init`` so each
DataLoader worker builds its own client connection (connections must not
be shared across fork).
"""
from __future__ import annotations

import io
import tarfile
from typing import Iterator, Optional

from curvine_amd.conf import ClusterConf


class _LazySyncFs:
    """Per-process SyncFs holder (safe across DataLoader fork workers)."""

    def __init__(self, conf: ClusterConf):
        self.conf = conf
        self._fs = None
        self._pid = None

    def get(self):
        import os

        from curvine_amd.client.filesystem import SyncFs
        if self._fs is None or self._pid != os.getpid():
            self._fs = SyncFs(self.conf)
            self._pid = os.getpid()
        return self._fs


class CurvineFileDataset:
    """Map-style dataset: one item per file under a directory."""

    def __init__(self, conf: ClusterConf, root: str,
                 transform=None):
        self.holder = _LazySyncFs(conf)
        self.root = root
        self.transform = transform
        fs = self.holder.get()
        self.paths = [s.path for s in fs.list_status(root) if not s.is_dir]

    def __len__(self) -> int:
        return len(self.paths)

    def __getitem__(self, idx: int):
        data = self.holder.get().read_file(self.paths[idx])
        return self.transform(data) if self.transform else data


try:
    from torch.utils.data import IterableDataset as _TorchIterable
except ImportError:  # pragma: no cover
    class _TorchIterable:  # type: ignore[no-redef]
        pass


class CurvineShardDataset(_TorchIterable):
    """Iterable dataset over tar shards (WebDataset layout): yields
    (name, bytes) per tar member, sharded across DataLoader workers."""

    def __init__(self, conf: ClusterConf, shard_paths: list[str],
                 transform=None):
        self.holder = _LazySyncFs(conf)
        self.shards = list(shard_paths)
        self.transform = transform

    def _worker_shards(self) -> list[str]:
        try:
            import torch.utils.data as tud
            info = tud.get_worker_info()
        except ImportError:
            info = None
        if info is None:
            return self.shards
        return self.shards[info.id::info.num_workers]

    def __iter__(self) -> Iterator:
        fs = self.holder.get()
        for shard in self._worker_shards():
            blob = fs.read_file(shard)
            with tarfile.open(fileobj=io.BytesIO(blob)) as tf:
                for member in tf:
                    if not member.isfile():
                        continue
                    payload = tf.extractfile(member).read()
                    item = (member.name, payload)
                    yield self.transform(item) if self.transform else item


def curvine_worker_init(_worker_id: int) -> None:
    """No-op placeholder (per-process clients are rebuilt lazily)."""
