"""Offset allocator for byte arenas (HBM / host-memory tiers).

Analog of the reference's `BdevOffsetAllocator`
(/root/reference/crates/adapters/curvine-storage-local/src/layout/
bdev_layout.rs:30-111, dir_state.rs:52): blocks live at offsets inside one
big arena, so "rename/finalize" is an epoch/generation map instead of a
file rename (SURVEY.md §7 hard-parts note).

First-fit free list with address-ordered coalescing; 4 KiB granularity.
Supports shrinking a live allocation (free the tail when a block finalizes
below its reserved size).
"""
from __future__ import annotations

import bisect

from curvine_amd import errors as err

ALIGN = 4096


def _align_up(n: int) -> int:
    return (n + ALIGN - 1) & ~(ALIGN - 1)


class ArenaAllocator:
    def __init__(self, capacity: int):
        self.capacity = capacity
        # free list: sorted list of [off, len]
        self.free: list[list[int]] = [[0, capacity]]
        self.used = 0
        self.allocs: dict[int, int] = {}   # off -> len

    def alloc(self, size: int) -> int:
        size = _align_up(max(size, ALIGN))
        for i, (off, ln) in enumerate(self.free):
            if ln >= size:
                if ln == size:
                    self.free.pop(i)
                else:
                    self.free[i] = [off + size, ln - size]
                self.allocs[off] = size
                self.used += size
                return off
        raise err.CapacityExceeded(
            f"arena: need {size}, used {self.used}/{self.capacity}")

    def free_range(self, off: int, ln: int) -> None:
        """Insert [off, ln) into the free list, coalescing neighbours."""
        i = bisect.bisect_left(self.free, [off, 0])
        self.free.insert(i, [off, ln])
        # coalesce with next
        if i + 1 < len(self.free) and self.free[i][0] + self.free[i][1] == self.free[i + 1][0]:
            self.free[i][1] += self.free[i + 1][1]
            self.free.pop(i + 1)
        # coalesce with prev
        if i > 0 and self.free[i - 1][0] + self.free[i - 1][1] == self.free[i][0]:
            self.free[i - 1][1] += self.free[i][1]
            self.free.pop(i)

    def release(self, off: int) -> int:
        ln = self.allocs.pop(off, None)
        if ln is None:
            return 0
        self.used -= ln
        self.free_range(off, ln)
        return ln

    def shrink(self, off: int, new_size: int) -> int:
        """Keep [off, new_size'), free the tail. Returns retained size."""
        ln = self.allocs.get(off)
        if ln is None:
            raise err.BlockNotFound(f"allocation at {off}")
        keep = _align_up(max(new_size, ALIGN))
        if keep >= ln:
            return ln
        self.allocs[off] = keep
        self.used -= ln - keep
        self.free_range(off + keep, ln - keep)
        return keep

    @property
    def available(self) -> int:
        return self.capacity - self.used
