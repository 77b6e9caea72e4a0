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
from typing import Iterator

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


# ---------------------------------------------------------------------------
# Zero-host-hop ingest: tar samples living in the HBM cache are gathered
# straight into torch device tensors (reader.pread_gather -> on-chip
# copy_extents_kernel).  This is the MI355X answer to BASELINE config[4]'s
# "2 TB WebDataset shards -> PyTorch DataLoader" path: indexing walks only
# the 512-byte tar headers through the local short-circuit reader; payload
# bytes never cross the host.
# ---------------------------------------------------------------------------

def index_tar(reader, length: int) -> list[tuple[str, int, int]]:
    """Walk ustar headers via pread: [(member_name, payload_off, size)].
    Only regular files are returned; other entry types are skipped by
    size.  GNU long-name entries ('L') apply to the next member."""
    out = []
    off = 0
    pending_name = None
    while off + 512 <= length:
        hdr = reader.pread(off, 512)
        if len(hdr) < 512 or hdr[0] == 0:
            break                      # end-of-archive zero block
        size = int(bytes(hdr[124:136]).split(b"\0")[0].strip() or b"0", 8)
        typ = hdr[156:157]
        payload = off + 512
        if typ == b"L":                # GNU longname: payload is the name
            pending_name = reader.pread(payload, size).rstrip(b"\0").decode()
        elif typ == b"x":              # pax extended header: path= record
            body = bytes(reader.pread(payload, size))
            pos = 0
            while pos < len(body):
                sp = body.index(b" ", pos)
                rec_len = int(body[pos:sp])
                rec = body[sp + 1:pos + rec_len - 1]   # strip trailing \n
                if rec.startswith(b"path="):
                    pending_name = rec[5:].decode()
                pos += rec_len
        elif typ in (b"0", b"\0"):
            name = pending_name or bytes(hdr[0:100]).split(b"\0")[0].decode()
            pending_name = None
            out.append((name, payload, size))
        elif typ != b"g":              # pax global header: keep pending
            pending_name = None
        off = payload + ((size + 511) & ~511)
    return out


class CurvineDeviceLoader:
    """Batched raw-sample ingest: yields (tensor, sections, names) per
    batch, where ``tensor`` is one uint8 tensor on ``device`` holding the
    concatenated payloads and ``sections`` is [(start, len)] per sample.

    HBM-tier shards + cuda device = pure D2D gather (no host hop); MEM
    tier + cpu device = host-side scatter memcpy (CPU-testable)."""

    def __init__(self, conf: ClusterConf, shard_paths: list[str],
                 device: str = "cuda:0", batch_size: int = 64,
                 shuffle: bool = False, seed: int = 0):
        from curvine_amd.client.filesystem import SyncFs
        from curvine_amd.client.reader import SyncLocalReader
        self.device = device
        self.batch_size = batch_size
        self.shuffle = shuffle
        self.seed = seed
        self._fs = SyncFs(conf)
        self._readers = []
        self._samples = []             # (reader_idx, name, off, size)
        self._plans = None             # cached batch gather plans
        for sp in shard_paths:
            fb = self._fs.call(self._fs.fs.client.open(sp))
            r = SyncLocalReader(fb)
            ridx = len(self._readers)
            self._readers.append(r)
            for name, off, size in index_tar(r, fb.status.length):
                self._samples.append((ridx, name, off, size))

    def __len__(self) -> int:
        return (len(self._samples) + self.batch_size - 1) // self.batch_size

    @property
    def num_samples(self) -> int:
        return len(self._samples)

    def _plan(self, on_dev: bool):
        """Batch plans with sample extents pre-resolved down to arena
        triples (valid while the readers stay open — they pin the
        blocks): iteration is just torch.empty + replayed gather calls.
        The gather kernel runs ~13 µs per 16 MiB batch; Python must not
        dominate it."""
        order = list(range(len(self._samples)))
        if self.shuffle:
            import random
            random.Random(self.seed).shuffle(order)
        plans = []
        bs = self.batch_size
        for b0 in range(0, len(order), bs):
            batch = [self._samples[i] for i in order[b0:b0 + bs]]
            per_reader: dict[int, list] = {}
            sections, names = [], []
            pos = 0
            for ridx, name, off, size in batch:
                per_reader.setdefault(ridx, []).append((off, size, pos))
                sections.append((pos, size))
                names.append(name)
                pos += size
            execs = []
            for ridx, samples in per_reader.items():
                groups, slow, _ = self._readers[ridx].resolve_gather(
                    samples, on_dev)
                execs.append((self._readers[ridx], groups, slow))
            plans.append((pos, execs, sections, names))
        return plans

    def __iter__(self):
        import torch
        dev = torch.device(self.device)
        on_dev = dev.type != "cpu"
        if self._plans is None:
            self._plans = self._plan(on_dev)
        for total, execs, sections, names in self._plans:
            out = torch.empty(total, dtype=torch.uint8, device=dev)
            dst = out.data_ptr()
            for reader, groups, slow in execs:
                reader.exec_gather(groups, slow, dst)
            if on_dev:
                torch.cuda.synchronize(dev)
            yield out, sections, names

    def close(self) -> None:
        for r in self._readers:
            r.close()
        self._fs.shutdown()
