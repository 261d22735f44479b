"""GPU data plane: HBM block pool served over ROCm IPC / xGMI.

The MI355X realization of the reference's registered-memory + one-sided
READ machinery:

* served map outputs live in hipMalloc **slabs** owned by the native lib,
  sub-allocated by the same buddy BlockPool as host mode. Each slab is
  exported once via hipIpcGetMemHandle; its 64-byte handle + size are
  published in a fixed slot of the owner's host *metadata segment*, so a
  fetcher resolves (key -> handle) one-sidedly — no RPC, mirroring how the
  reference smuggles rkeys through one-sided table reads.
* a fetch is a batch of hipMemcpyAsync D2D on a per-peer-GPU stream with
  one completion event — the scatter-list RDMA READ with a single signaled
  WR (RdmaChannel.java:484-517). ``issue_read_into`` is non-blocking
  (enqueue + event id); the fetcher's completion thread polls events and
  dispatches results — the event-driven CQ analog (RdmaChannel.java:
  683-870, RdmaThread.java:45-58). Blocking ``read_device`` forms remain
  for simple callers.
* ``prebuild`` pre-opens peer state in the background the moment the
  driver announces membership — the reference pre-builds its channel mesh
  the same way (RdmaShuffleManager.scala:121-126) so reduce stages never
  pay first-touch connection latency.
* GPU segment ids carry the high bit (0x8000); the low 15 bits index the
  owner's slab-table slot. Addresses in BlockLocations are slab OFFSETS,
  valid in any importing process.

Fetch destinations land in a caller-reusable device *arena* (each
coalesced fetch pre-assigned an offset — reader.py), or in transient
torch tensors; only *served* memory needs the IPC-exported slabs.
"""

from __future__ import annotations

import logging
import struct
import threading
import time
from typing import Dict, Optional

import torch

from .block_pool import BlockPool
from .map_output import make_key, split_key
from .segments import META_SEGMENT_ID, SLAB_SLOT_SIZE, SLAB_TABLE_MAX

log = logging.getLogger(__name__)

GPU_SEG_FLAG = 0x8000


def is_gpu_key(key: int) -> bool:
    return bool((key & 0xFFFF) & GPU_SEG_FLAG)


class GpuDataPlane:
    def __init__(self, conf, executor_id: int, meta_segment, registry,
                 device: Optional[int] = None, peer_device=None):
        from .ops import load
        self.hs = load()
        self.conf = conf
        self.executor_id = executor_id
        self.meta_segment = meta_segment     # our own (HostSegment)
        self.registry = registry             # peers' segments (SegmentRegistry)
        self.device = conf.resolved_gpu_id() if device is None else device
        # exec_id -> peer GPU ordinal (the xGMI link selector for copy
        # streams); installed by the manager from announce membership
        self._peer_device = peer_device or (lambda eid: self.device)
        if torch.cuda.is_available():
            torch.cuda.set_device(self.device)
        self.hs.set_device(self.device)
        self._slab_ids: Dict[int, int] = {}   # slot -> native slab id
        self._slab_bases: Dict[int, int] = {}  # slot -> local base ptr
        self._next_slot = 0
        self._free_slots: list = []            # recycled slot indices
        self._slab_gens: Dict[int, int] = {}   # slot -> generation
        # remote key -> (imported base, generation). Slots RECYCLE, so a
        # cached mapping is valid only while the published generation
        # matches; resolve() re-checks the 72-byte slot entry per fetch
        # (a ~1 us pread against >=100 us copies)
        self._peer_bases: Dict[int, tuple] = {}
        self._lock = threading.Lock()
        self._key_locks: Dict[int, threading.Lock] = {}  # per-key resolve
        pool_max = conf.hbm_pool_size or self._auto_pool_bytes()
        # slab-table slots are finite: grow the slab size until the whole
        # pool fits the table (so a 288 GB pool is actually servable even
        # with a small configured slab size — VERDICT r01)
        slab_size = conf.hbm_slab_size
        while slab_size * (SLAB_TABLE_MAX - 8) < pool_max:
            slab_size *= 2
        if slab_size != conf.hbm_slab_size:
            log.info("hbm slab size auto-raised %d -> %d for %d-byte pool",
                     conf.hbm_slab_size, slab_size, pool_max)
        self.pool = BlockPool(
            slab_size=slab_size, max_bytes=pool_max,
            alloc_slab=self._alloc_slab, free_slab=self._free_slab)

    def _auto_pool_bytes(self) -> int:
        if torch.cuda.is_available():
            free, _total = torch.cuda.mem_get_info(self.device)
            return max(self.conf.hbm_slab_size, int(free * 0.5))
        return 4 << 30

    # ------------------------------------------------------------------
    # slab backend for BlockPool

    SIZE_MASK = (1 << 48) - 1   # slot entry: size (48b) | generation (16b)

    def _alloc_slab(self, size: int) -> int:
        with self._lock:
            if self._free_slots:
                slot = self._free_slots.pop()
            else:
                slot = self._next_slot
                if slot >= SLAB_TABLE_MAX:
                    raise MemoryError("slab table full")
                self._next_slot += 1
            gen = (self._slab_gens.get(slot, 0) + 1) & 0xFFFF
            self._slab_gens[slot] = gen
        try:
            sid = self.hs.slab_alloc(size)
        except RuntimeError:
            # torch's caching allocator may be hoarding freed blocks;
            # release them and retry once before giving up
            if torch.cuda.is_available():
                torch.cuda.empty_cache()
            sid = self.hs.slab_alloc(size)
        self._slab_ids[slot] = sid
        self._slab_bases[slot] = self.hs.slab_base(sid)
        handle = self.hs.slab_handle(sid)
        # publish (gen<<48 | size, handle) one-sidedly; the generation
        # lets importers detect RECYCLED slots and reopen (liveness
        # guarantees no in-flight fetch targets the old slab)
        self.meta_segment.write(
            16 + slot * SLAB_SLOT_SIZE,
            struct.pack("<Q", size | (gen << 48)) + handle)
        return GPU_SEG_FLAG | slot

    def _free_slab(self, seg_id: int) -> None:
        slot = seg_id & 0x7FFF
        self.meta_segment.write(16 + slot * SLAB_SLOT_SIZE, b"\0" * 8)
        sid = self._slab_ids.pop(slot)
        self._slab_bases.pop(slot)
        self.hs.slab_free(sid)
        with self._lock:
            self._free_slots.append(slot)   # slot recycles (gen advances)

    def local_base(self, seg_id: int) -> int:
        return self._slab_bases[seg_id & 0x7FFF]

    # ------------------------------------------------------------------
    # mesh pre-build (reference RdmaShuffleManager.scala:121-126)

    def prebuild(self, members, deadline_s: float = 5.0) -> None:
        """One-shot per-peer setup on Announce: enable xGMI peer access to
        every peer GPU and open peers' metadata segments. Peers whose
        segments are not up yet are retried until ``deadline_s``; whatever
        stays unopened resolves lazily on first fetch — pre-build is an
        optimization, never a correctness gate. Slab handles themselves
        are imported continuously by ``importer_tick`` (slabs appear as
        writers commit, long after Announce)."""
        # background thread: pin OUR device first — peer-access enables
        # and IPC opens bind to the CURRENT device context, and threads
        # default to device 0 (wrong context on executors using GPU 1-7)
        self.hs.set_device(self.device)
        todo = {m.executor_id: m for m in members
                if m.executor_id != self.executor_id}
        deadline = time.monotonic() + deadline_s
        while todo and time.monotonic() < deadline:
            for eid, m in list(todo.items()):
                try:
                    if torch.cuda.is_available() and m.gpu_id >= 0 \
                            and m.gpu_id != self.device:
                        self.hs.enable_peer_access(m.gpu_id)
                    meta_key = make_key(eid, META_SEGMENT_ID)
                    self.registry.reader(meta_key)   # open + cache the fd
                    self._import_published_slabs(eid)
                    todo.pop(eid)
                except FileNotFoundError:
                    continue   # peer not initialized yet; retry
                except Exception as e:   # pragma: no cover - defensive
                    log.warning("prebuild for peer %d failed: %s", eid, e)
                    todo.pop(eid)
            if todo:
                time.sleep(0.02)

    def importer_tick(self, members) -> None:
        """One sweep of every same-host peer's slab table, importing any
        newly published handles — keeps hop-3 issue latency independent of
        first-touch (the reference needs no analog: ibverbs rkeys are
        usable without a per-MR open; hipIpc handles are not)."""
        self.hs.set_device(self.device)   # importer thread: our context
        for m in members:
            if m.executor_id == self.executor_id:
                continue
            try:
                self._import_published_slabs(m.executor_id)
            except FileNotFoundError:
                pass   # peer not initialized yet

    def _import_published_slabs(self, exec_id: int) -> None:
        """Open every slab handle the peer has already published (one
        sweep of its slab table). Later slabs import lazily on fetch."""
        if not torch.cuda.is_available():
            return
        raw = self.registry.read(make_key(exec_id, META_SEGMENT_ID),
                                 16, SLAB_TABLE_MAX * SLAB_SLOT_SIZE)
        for slot in range(SLAB_TABLE_MAX):
            off = slot * SLAB_SLOT_SIZE
            (entry,) = struct.unpack_from("<Q", raw, off)
            size, gen = entry & self.SIZE_MASK, entry >> 48
            if size == 0:
                continue
            key = make_key(exec_id, GPU_SEG_FLAG | slot)
            cached = self._peer_bases.get(key)
            if cached is not None and cached[1] == gen:
                continue
            with self._key_lock(key):
                cached = self._peer_bases.get(key)
                if cached is not None and cached[1] == gen:
                    continue
                try:
                    base = self.hs.ipc_open(bytes(raw[off + 8:off + 8 + 64]))
                except RuntimeError:
                    continue   # peer freed it between read and open
                if cached is not None:
                    try:
                        self.hs.ipc_close(cached[0])
                    except Exception:
                        pass
                with self._lock:
                    self._peer_bases[key] = (base, gen)

    # ------------------------------------------------------------------
    # one-sided fetch (hop 3)

    def _key_lock(self, key: int) -> threading.Lock:
        with self._lock:
            lk = self._key_locks.get(key)
            if lk is None:
                lk = self._key_locks[key] = threading.Lock()
            return lk

    def resolve(self, key: int) -> int:
        """key -> device pointer of the owning slab in THIS process.
        Opens the peer's IPC handle on first touch and caches it; the
        published GENERATION is re-checked each resolve (slots recycle —
        a ~1 us pread guards against serving a reused slot through a
        stale mapping). Per-key locks so resolving different slabs never
        serializes."""
        exec_id, seg_id = split_key(key)
        slot = seg_id & 0x7FFF
        if exec_id == self.executor_id:
            return self._slab_bases[slot]
        raw = self.registry.read(
            make_key(exec_id, META_SEGMENT_ID),
            16 + slot * SLAB_SLOT_SIZE, SLAB_SLOT_SIZE)
        (entry,) = struct.unpack_from("<Q", raw, 0)
        size, gen = entry & self.SIZE_MASK, entry >> 48
        if size == 0:
            raise RuntimeError(
                f"peer {exec_id} slab slot {slot} not published")
        cached = self._peer_bases.get(key)
        if cached is not None and cached[1] == gen:
            return cached[0]
        with self._key_lock(key):
            cached = self._peer_bases.get(key)
            if cached is not None and cached[1] == gen:
                return cached[0]
            # hop-2 worker threads reach here: the hipIpcOpenMemHandle
            # below must run in OUR device context, not thread-default 0
            self.hs.set_device(self.device)
            base = self.hs.ipc_open(bytes(raw[8:8 + 64]))
            if cached is not None:   # recycled slot: drop the old mapping
                try:
                    self.hs.ipc_close(cached[0])
                except Exception:
                    pass
            with self._lock:
                self._peer_bases[key] = (base, gen)
            return base

    # kept under the old name for callers/tests of r01
    _resolve_base = resolve

    def issue_read_into(self, key: int, addr: int, length: int,
                        dst_ptr: int) -> int:
        """NON-BLOCKING one-sided read: enqueue the xGMI copy on the
        owning peer GPU's stream, return the completion event id. The
        caller (the fetcher's completion thread) polls ``poll_event`` —
        in-flight depth is bounded by the fetcher's per-peer caps, not by
        blocked threads."""
        exec_id, _ = split_key(key)
        self.hs.set_device(self.device)
        base = self.resolve(key)
        peer = self._stream_slot(exec_id)
        return self.hs.read_batch(peer, [dst_ptr], [base + addr], [length])

    def _stream_slot(self, exec_id: int) -> int:
        """Copy-stream selector: the peer's GPU ordinal — streams map to
        xGMI links, not executor ids."""
        if exec_id == self.executor_id:
            return self.device % 64
        try:
            return int(self._peer_device(exec_id)) % 64
        except Exception:
            return exec_id % 64

    def poll_event(self, ev: int) -> bool:
        return self.hs.poll_event(ev)

    def wait_event(self, ev: int) -> None:
        self.hs.wait_event(ev)

    def read_device(self, key: int, addr: int, length: int) -> torch.Tensor:
        """Blocking one-sided read into a fresh device tensor (simple
        callers; the fetcher uses issue_read_into + completion thread)."""
        dst = torch.empty(length, dtype=torch.uint8,
                          device=f"cuda:{self.device}")
        ev = self.issue_read_into(key, addr, length, dst.data_ptr())
        self.hs.wait_event(ev)
        return dst

    def read_device_into(self, key: int, addr: int, length: int,
                         dst_ptr: int) -> None:
        """Blocking pre-placed one-sided read (arena slow path)."""
        ev = self.issue_read_into(key, addr, length, dst_ptr)
        self.hs.wait_event(ev)

    def stop(self) -> None:
        for key, (base, _gen) in list(self._peer_bases.items()):
            exec_id, _ = split_key(key)
            if exec_id != self.executor_id:
                try:
                    self.hs.ipc_close(base)
                except Exception:
                    pass
        self._peer_bases.clear()
        for slot, sid in list(self._slab_ids.items()):
            try:
                self.hs.slab_free(sid)
            except Exception:
                pass
        self._slab_ids.clear()
        self._slab_bases.clear()
