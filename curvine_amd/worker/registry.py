"""In-process worker registry for short-circuit I/O.

The MI355X deployment colocates the FUSE/client process with the worker
owning the local GPU's HBM arena (one process per GPU).  When the target
worker lives in this process, clients bypass RPC entirely and read/write
the block store directly — the embedded analog of the reference's
short-circuit path (block_store.rs:253-271, block_reader_local.rs), which
discloses a file path; an HBM extent cannot cross processes without
dmabuf IPC, so colocated-in-process is the designed-for fast path.
"""
from __future__ import annotations

import os

_stores: dict[int, "object"] = {}
_pids: dict[int, int] = {}


def register(worker_id: int, store) -> None:
    _stores[worker_id] = store
    _pids[worker_id] = os.getpid()


def unregister(worker_id: int) -> None:
    _stores.pop(worker_id, None)
    _pids.pop(worker_id, None)


def lookup(worker_id: int):
    """None after fork: a child inherits the registry dict but not usable
    HIP contexts/arenas — forked DataLoader workers must go remote."""
    if _pids.get(worker_id) != os.getpid():
        return None
    return _stores.get(worker_id)
