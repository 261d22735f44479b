"""Pooled block allocator over large slabs — host-shm or HBM.

MI355X re-design of the reference's registered-buffer pool
(RdmaBufferManager.java:93-211): the reference rounds requests to
powers of two with a 16 KiB minimum block, keeps per-size free stacks, and
LRU-trims idle memory past a watermark. Here the same size-class policy is
implemented as a **buddy allocator** per slab, so freed blocks of different
classes actually coalesce and the pool can satisfy any mix of sizes from a
fixed HBM budget (288 GB/GPU is large but finite, and HBM cannot be grown
the way pinned host memory could be re-registered).

The allocator is pure control-plane bookkeeping (offsets into slabs); the
slabs themselves are provided by a backend:

* host mode: one /dev/shm segment per slab (segments.HostSegment)
* GPU mode: one hipMalloc per slab, exported once via hipIpcGetMemHandle
  (ops/hipshuffle). Allocation decisions stay here, in tested Python —
  per-block (~8 MiB) rates make this cheap.
"""

from __future__ import annotations

import threading
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional

MIN_BLOCK = 16 << 10   # reference MIN_BLOCK_SIZE (RdmaBufferManager.java:93)


def _round_pow2(n: int) -> int:
    if n <= MIN_BLOCK:
        return MIN_BLOCK
    return 1 << (n - 1).bit_length()


@dataclass
class Block:
    """One allocation: (segment_id, offset, capacity); ``length`` is the
    caller's used size. Refcounted like RdmaRegisteredBuffer.java:45-62."""
    segment_id: int
    offset: int
    capacity: int
    pool: "BlockPool"
    refcount: int = 1

    def retain(self) -> "Block":
        with self.pool._lock:
            self.refcount += 1
        return self

    def release(self) -> None:
        # refcount ops share the pool lock (the reference uses an atomic,
        # RdmaRegisteredBuffer.java:45-62); put() re-takes it, so drop first
        with self.pool._lock:
            self.refcount -= 1
            if self.refcount > 0:
                return
        self.pool.put(self)


class _BuddySlab:
    """Classic buddy allocator over one slab of pow-2 size."""

    def __init__(self, segment_id: int, size: int):
        if size & (size - 1):
            raise ValueError("slab size must be a power of two")
        self.segment_id = segment_id
        self.size = size
        self.max_order = (size // MIN_BLOCK).bit_length() - 1
        # free lists per order; order o block size = MIN_BLOCK << o
        self.free: List[set] = [set() for _ in range(self.max_order + 1)]
        self.free[self.max_order].add(0)
        self.allocated: Dict[int, int] = {}  # offset -> order
        self.used_bytes = 0

    def alloc(self, size: int) -> Optional[int]:
        order = max(0, (max(size, MIN_BLOCK) // MIN_BLOCK - 1).bit_length())
        if (MIN_BLOCK << order) < size:
            order += 1
        if order > self.max_order:
            return None
        o = order
        while o <= self.max_order and not self.free[o]:
            o += 1
        if o > self.max_order:
            return None
        off = self.free[o].pop()
        while o > order:  # split down
            o -= 1
            buddy = off + (MIN_BLOCK << o)
            self.free[o].add(buddy)
        self.allocated[off] = order
        self.used_bytes += MIN_BLOCK << order
        return off

    def dealloc(self, offset: int) -> None:
        order = self.allocated.pop(offset)
        self.used_bytes -= MIN_BLOCK << order
        while order < self.max_order:
            buddy = offset ^ (MIN_BLOCK << order)
            if buddy not in self.free[order]:
                break
            self.free[order].discard(buddy)
            offset = min(offset, buddy)
            order += 1
        self.free[order].add(offset)

    @property
    def fully_free(self) -> bool:
        return self.used_bytes == 0


@dataclass
class PoolStats:
    allocs: int = 0
    frees: int = 0
    slab_count: int = 0
    slab_bytes: int = 0
    used_bytes: int = 0
    alloc_by_class: Dict[int, int] = field(default_factory=dict)


class BlockPool:
    """Multi-slab buddy pool with lazy slab growth and idle-slab trimming.

    * ``get(size)``: pow-2-rounded allocation (reference
      RdmaBufferManager.java:147-161); grows by one slab when all slabs are
      exhausted, up to ``max_bytes``.
    * ``put(block)``: return + coalesce; when total slab bytes exceed
      ``trim_high`` × max and a slab is fully free, the slab is released
      back to the backend — the reference's LRU cleaner semantics
      (RdmaBufferManager.java:163-211) expressed at slab granularity.
    """

    def __init__(self, slab_size: int, max_bytes: int,
                 alloc_slab: Callable[[int], int],
                 free_slab: Callable[[int], None] = None,
                 trim_high: float = 0.9, trim_low: float = 0.65):
        if slab_size & (slab_size - 1):
            raise ValueError("slab size must be a power of two")
        self.slab_size = slab_size
        self.max_bytes = max_bytes
        self._alloc_slab = alloc_slab
        self._free_slab = free_slab or (lambda seg_id: None)
        self.trim_high = trim_high
        self.trim_low = trim_low
        self._slabs: Dict[int, _BuddySlab] = {}
        self._lock = threading.Lock()
        self.stats = PoolStats()

    def get(self, size: int) -> Block:
        if size <= 0:
            raise ValueError("size must be positive")
        rounded = _round_pow2(size)
        if rounded > self.slab_size:
            raise MemoryError(
                f"request {size} exceeds slab size {self.slab_size}")
        with self._lock:
            for slab in self._slabs.values():
                off = slab.alloc(rounded)
                if off is not None:
                    return self._record(slab, off, rounded)
            if self.stats.slab_bytes + self.slab_size > self.max_bytes:
                raise MemoryError(
                    f"pool exhausted: {self.stats.slab_bytes} + slab > {self.max_bytes}")
            seg_id = self._alloc_slab(self.slab_size)
            slab = _BuddySlab(seg_id, self.slab_size)
            self._slabs[seg_id] = slab
            self.stats.slab_count += 1
            self.stats.slab_bytes += self.slab_size
            off = slab.alloc(rounded)
            assert off is not None
            return self._record(slab, off, rounded)

    def _record(self, slab: _BuddySlab, off: int, rounded: int) -> Block:
        self.stats.allocs += 1
        self.stats.used_bytes += rounded
        self.stats.alloc_by_class[rounded] = self.stats.alloc_by_class.get(rounded, 0) + 1
        return Block(slab.segment_id, off, rounded, self)

    def put(self, block: Block) -> None:
        with self._lock:
            slab = self._slabs[block.segment_id]
            slab.dealloc(block.offset)
            self.stats.frees += 1
            self.stats.used_bytes -= block.capacity
            self._maybe_trim()

    def _maybe_trim(self) -> None:
        if self.stats.slab_bytes <= self.trim_high * self.max_bytes:
            return
        target = int(self.trim_low * self.max_bytes)
        for seg_id in [s for s, sl in self._slabs.items() if sl.fully_free]:
            if self.stats.slab_bytes <= target:
                break
            del self._slabs[seg_id]
            self.stats.slab_count -= 1
            self.stats.slab_bytes -= self.slab_size
            self._free_slab(seg_id)

    def preallocate(self, size: int, count: int) -> None:
        """Warm the pool (reference preAllocate, RdmaBufferManager.java:124-135)."""
        blocks = [self.get(size) for _ in range(count)]
        for b in blocks:
            b.release()

    @property
    def idle_bytes(self) -> int:
        return self.stats.slab_bytes - self.stats.used_bytes

    def format_stats(self) -> str:
        s = self.stats
        classes = ", ".join(f"{k >> 10}k:{v}" for k, v in sorted(s.alloc_by_class.items()))
        return (f"pool: slabs={s.slab_count} ({s.slab_bytes >> 20} MiB) "
                f"used={s.used_bytes >> 20} MiB allocs={s.allocs} frees={s.frees} "
                f"by_class=[{classes}]")
