"""Shuffle read path — the 3-hop one-sided fetch pipeline.

Re-design of RdmaShuffleReader.scala + RdmaShuffleFetcherIterator.scala
(the reference's hot path). Same structure:

* HOP 1: read the whole driver table once, cached per shuffle
  (RdmaShuffleManager.scala:341-376) — here a polled shm read (driver RPC
  lane across hosts).
* HOP 2: per map task, one-sided read of its MapTaskOutput entries for
  [start_partition, end_partition] (RdmaShuffleFetcherIterator.scala:
  297-311). Issued ASYNC, grouped per source executor, through the fetch
  pool — never on the constructing thread's critical path (the reference
  runs hop 2 inside table-read onComplete callbacks the same way).
* HOP 3: coalesce consecutive block locations up to shuffle_read_block_size
  (:240-263), gate on max_bytes_in_flight (:264-273) and a per-SOURCE
  request cap (:82-83 — per channel in the reference; per source executor
  here, so one hot peer's xGMI link cannot starve the others), randomize
  the pending queue to spread load across source executors (:74-79), then
  issue one-sided data reads; results surface through a blocking queue as
  they land (:340-382).

Completion model (event-driven, RdmaChannel.java:683-870 +
RdmaThread.java:45-58): GPU one-sided reads are ENQUEUED (hipMemcpyAsync
batch + recorded event = the signaled last WR) and a single completion
thread polls the in-flight events, dispatching results — in-flight depth
is set by the flow-control caps, not by a count of blocked threads. Host
segment preads and the cross-host TCP lane still run on the small thread
pool (they are genuinely blocking OS calls).
"""

from __future__ import annotations

import os
import queue
import random
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from dataclasses import dataclass
from typing import Iterator, List, Optional

from .manager import ShuffleHandle, ShuffleManager
from .map_output import ENTRY_SIZE, MapTaskOutput, split_key
from .stats import TaskMetrics


@dataclass(frozen=True)
class BlockRef:
    """One shuffle block to fetch: (map_id, partition) at (key, addr, len)."""
    map_id: int
    partition: int
    key: int
    addr: int
    length: int


@dataclass
class CoalescedFetch:
    """One one-sided read covering >=1 consecutive blocks (same segment,
    contiguous addresses) — the reference's scatter-list READ collapses to
    a single contiguous read here because consecutive partitions of one map
    task are laid out back-to-back by the writer."""
    key: int
    addr: int
    length: int
    blocks: List[BlockRef]
    arena_off: int = -1          # destination offset in the fetch arena


@dataclass
class FetchResult:
    fetch: Optional[CoalescedFetch]
    data: Optional[bytes]
    error: Optional[BaseException] = None
    latency_ms: float = 0.0


_SENTINEL = FetchResult(None, None)   # wake-up after async hop-2 drains


class FetchFailedError(RuntimeError):
    """Analog of Spark's FetchFailedException — one failure fails the task
    (reference RdmaShuffleFetcherIterator.scala:167)."""


def parse_cpu_list(spec: str) -> list:
    """Parse the reference's cpuList format '0-3,8,10-11'
    (RdmaShuffleConf.scala:89) into a CPU id list for fetch-thread
    affinity (the CQ-poller placement analog, RdmaThread.java:45-47)."""
    cpus = []
    for part in (spec or "").split(","):
        part = part.strip()
        if not part:
            continue
        if "-" in part:
            a, b = part.split("-")
            cpus.extend(range(int(a), int(b) + 1))
        else:
            cpus.append(int(part))
    return cpus


def coalesce_blocks(blocks: List[BlockRef], max_bytes: int,
                    max_blocks: int = 1 << 30) -> List[CoalescedFetch]:
    """Group blocks into minimal one-sided reads.

    Blocks merge while they are in the same segment AND contiguous AND the
    running size stays under max_bytes (reference :240-263; a block larger
    than max_bytes still fetches as one read). Zero-length blocks are
    dropped.
    """
    out: List[CoalescedFetch] = []
    cur: Optional[CoalescedFetch] = None
    for b in blocks:
        if b.length == 0:
            continue
        if (cur is not None and b.key == cur.key
                and b.addr == cur.addr + cur.length
                and cur.length + b.length <= max_bytes
                and len(cur.blocks) < max_blocks):
            cur.blocks.append(b)
            cur.length += b.length
        else:
            cur = CoalescedFetch(b.key, b.addr, b.length, [b])
            out.append(cur)
    return out


class FetcherIterator:
    """Async fetch pipeline with byte-budget + per-source flow control."""

    def __init__(self, manager: ShuffleManager, handle: ShuffleHandle,
                 start_partition: int, end_partition: int,
                 num_workers: int = 8, seed: Optional[int] = None,
                 arena: Optional[object] = None):
        self.manager = manager
        self.handle = handle
        self.start_partition = start_partition
        self.end_partition = end_partition  # inclusive
        self.metrics = TaskMetrics()
        conf = manager.conf
        self._cpus = parse_cpu_list(conf.cpu_list)
        self._max_bytes_in_flight = conf.max_bytes_in_flight
        self._read_block = conf.shuffle_read_block_size
        # per-SOURCE outstanding-read cap (reference caps per channel:
        # RdmaShuffleFetcherIterator.scala:82-83)
        self._per_source_limit = conf.resolved_read_requests_limit()
        self._results: "queue.Queue[FetchResult]" = queue.Queue()
        self._pending: List[CoalescedFetch] = []
        self._lock = threading.Lock()
        self._bytes_in_flight = 0
        self._reqs_by_source: dict = {}
        self._outstanding = 0     # fetches created, not yet surfaced
        self._hop2_pending = 1    # +1 sentinel for the submit phase itself
        self._pool = ThreadPoolExecutor(max_workers=num_workers,
                                        thread_name_prefix="sparkrdma-fetch",
                                        initializer=self._pin_worker)
        self._rng = random.Random(seed)
        self._failed: Optional[BaseException] = None
        self._done = threading.Event()
        # fetch arena: one device buffer all GPU one-sided reads land in at
        # pre-assigned offsets (bump-allocated as hop-2 results arrive)
        self._arena_buf = arena            # caller-provided hint (optional)
        self._arena_used = 0
        self._deferred: List[CoalescedFetch] = []  # awaiting arena alloc
        self.arena = None                  # exact-length view, set when known
        self.arena_complete = False
        self._inflight_gpu: dict = {}      # event id -> (fetch, t0)
        self._blocking_arena_inflight = 0  # pool fetches writing the arena
        self._paused = False               # arena-growth issue gate
        self._completion_thread: Optional[threading.Thread] = None
        self._pool.submit(self._start_async)

    # ------------------------------------------------------------------

    def _pin_worker(self) -> None:
        """Spread fetch workers over conf.cpuList (poller affinity parity,
        RdmaNode.java:222-279). No-op when the list is empty."""
        if self._cpus:
            try:
                os.sched_setaffinity(0, set(self._cpus))
            except OSError:
                pass

    # ------------------------------ hop 1 + 2 (async) ------------------

    def _start_async(self) -> None:
        """Runs on the pool: hop 1, then one hop-2 task per source
        executor. The constructor returns immediately; the consumer blocks
        on the results queue until data lands."""
        try:
            mgr = self.manager
            entries = mgr.get_map_task_output_table(self.handle)  # HOP 1
            groups: dict = {}   # owner exec -> [(map_id, table_addr, table_key)]
            for map_id, (table_addr, table_key) in enumerate(entries):
                owner = split_key(table_key)[0]
                groups.setdefault(owner, []).append(
                    (map_id, table_addr, table_key))
            with self._lock:
                self._hop2_pending += len(groups)
            for owner, maps in groups.items():
                self._pool.submit(self._hop2_group, owner, maps)
        except BaseException as e:
            self._results.put(FetchResult(None, None, error=e))
        finally:
            self._hop2_done()

    def _hop2_group(self, owner: int, maps: list) -> None:
        """HOP 2 for one source executor: read each of its map tasks'
        location entries (one-sided pread same-host, TCP lane across
        hosts — ADVICE r01), build + coalesce blocks, pre-resolve the
        slabs they point into, then enqueue the fetches."""
        mgr = self.manager
        span = self.end_partition - self.start_partition + 1
        try:
            blocks: List[BlockRef] = []
            remote_host = mgr.is_remote_host(owner)
            for map_id, table_addr, table_key in maps:
                addr = table_addr + self.start_partition * ENTRY_SIZE
                raw = (mgr.tcp_read(owner, table_key, addr, span * ENTRY_SIZE)
                       if remote_host
                       else mgr.remote_read(table_key, addr, span * ENTRY_SIZE))
                for i, loc in enumerate(MapTaskOutput.parse_locations(raw)):
                    blocks.append(BlockRef(map_id, self.start_partition + i,
                                           loc.key, loc.addr, loc.length))
            fetches = coalesce_blocks(blocks, self._read_block)
            # pre-resolve slab bases OFF the issue path (first IPC open of
            # a peer slab costs ms; pump()'s issue is then pure enqueue)
            if mgr.gpu is not None and not remote_host:
                from .gpu_plane import is_gpu_key
                for key in {f.key for f in fetches if is_gpu_key(f.key)}:
                    mgr.gpu.resolve(key)
            self._ingest(fetches)
        except BaseException as e:
            self._results.put(FetchResult(None, None, error=e))
        finally:
            self._hop2_done()

    def _hop2_done(self) -> None:
        with self._lock:
            self._hop2_pending -= 1
            finished = self._hop2_pending == 0
        if finished:
            self._finalize_arena()
            self._results.put(_SENTINEL)
        self._pump()

    def _ingest(self, fetches: List[CoalescedFetch]) -> None:
        """Register new fetches: assign arena destinations, randomize, add
        to the pending queue (reference randomized queue :74-79)."""
        use_arena = self.manager.gpu is not None
        with self._lock:
            if use_arena:
                for f in fetches:
                    if (self._arena_buf is not None
                            and self._arena_used + f.length
                            <= self._arena_buf.numel()):
                        f.arena_off = self._arena_used
                        self._arena_used += f.length
                    else:
                        # no hint (or hint exhausted): defer until all
                        # hop-2 results are in, then allocate exactly
                        self._deferred.append(f)
            self._rng.shuffle(fetches)
            self._pending.extend(f for f in fetches
                                 if not (use_arena and f.arena_off < 0))
            self._outstanding += len(fetches)

    def _finalize_arena(self) -> None:
        """All hop-2 results are in: place any deferred fetches (no-hint
        mode allocates the exact total now) and expose the arena view."""
        with self._lock:
            deferred, self._deferred = self._deferred, []
        if deferred:
            import torch
            total = sum(f.length for f in deferred)
            dev = f"cuda:{self.manager.gpu.device}"
            if self._arena_buf is None and self._arena_used == 0:
                self._arena_buf = torch.empty(total, dtype=torch.uint8,
                                              device=dev)
            else:
                # hint exhausted mid-stream: pause issuing, drain landed
                # writes, grow into a fresh buffer, copy the prefix
                with self._lock:
                    self._paused = True
                old, used = self._arena_buf, self._arena_used
                new = torch.empty(used + total, dtype=torch.uint8, device=dev)
                self._drain_inflight()
                if used:
                    new[:used] = old[:used]
                    torch.cuda.synchronize()
                self._arena_buf = new
                with self._lock:
                    self._paused = False
            with self._lock:
                for f in deferred:
                    f.arena_off = self._arena_used
                    self._arena_used += f.length
                self._pending.extend(deferred)
        if self.manager.gpu is not None and self._arena_buf is not None:
            self.arena = self._arena_buf[:self._arena_used]
        self.arena_complete = True

    def _drain_inflight(self) -> None:
        """Wait for every in-progress arena write to land (arena
        re-allocation barrier; rare — only when a caller's arena hint was
        undersized)."""
        while True:
            with self._lock:
                evs = list(self._inflight_gpu.keys())
                blocking = self._blocking_arena_inflight
            if not evs and blocking == 0:
                return
            for ev in evs:
                self.manager.gpu.wait_event(ev)
            time.sleep(0.0002)

    # ------------------------------ hop 3 -------------------------------

    def _pump(self) -> None:
        """Issue pending fetches while under the byte budget (reference
        :264-273, re-pumped from next() :365-374) and each source's
        outstanding-request cap (:82-83)."""
        while True:
            to_issue = []
            with self._lock:
                if self._failed is not None or self._paused:
                    return
                i = len(self._pending) - 1
                while i >= 0:
                    f = self._pending[i]
                    if (self._bytes_in_flight > 0
                            and self._bytes_in_flight + f.length
                            > self._max_bytes_in_flight):
                        break   # global byte budget: stop issuing
                    src = split_key(f.key)[0]
                    if (self._reqs_by_source.get(src, 0)
                            >= self._per_source_limit):
                        i -= 1   # this source saturated; try others
                        continue
                    self._pending.pop(i)
                    self._bytes_in_flight += f.length
                    self._reqs_by_source[src] = \
                        self._reqs_by_source.get(src, 0) + 1
                    to_issue.append(f)
                    i -= 1
            if not to_issue:
                return
            for f in to_issue:
                self._issue(f)

    def _issue(self, f: CoalescedFetch) -> None:
        from .gpu_plane import is_gpu_key
        mgr = self.manager
        owner = split_key(f.key)[0]
        if (f.arena_off >= 0 and is_gpu_key(f.key)
                and not mgr.is_remote_host(owner)):
            # event-driven lane: enqueue the xGMI copy; the completion
            # thread polls the event (no blocked thread per fetch)
            try:
                ev = mgr.gpu.issue_read_into(
                    f.key, f.addr, f.length,
                    self._arena_buf.data_ptr() + f.arena_off)
            except BaseException as e:
                self._results.put(FetchResult(f, None, error=e))
                return
            with self._lock:
                self._inflight_gpu[ev] = (f, time.perf_counter())
            self._ensure_completion_thread()
        else:
            if f.arena_off >= 0:
                with self._lock:
                    self._blocking_arena_inflight += 1
            self._pool.submit(self._do_fetch_blocking, f)

    def _ensure_completion_thread(self) -> None:
        if self._completion_thread is None:
            t = threading.Thread(target=self._completion_loop,
                                 name="sparkrdma-completion", daemon=True)
            self._completion_thread = t
            t.start()

    def _completion_loop(self) -> None:
        """The CQ-poller analog (RdmaThread.java:45-58): poll in-flight
        copy events, dispatch results as they complete."""
        if self._cpus:
            try:
                os.sched_setaffinity(0, set(self._cpus))
            except OSError:
                pass
        gpu = self.manager.gpu
        while not self._done.is_set():
            with self._lock:
                items = list(self._inflight_gpu.items())
            if not items:
                time.sleep(0.0002)
                continue
            completed = []
            for ev, (f, t0) in items:
                try:
                    if gpu.poll_event(ev):
                        completed.append((ev, f, t0, None))
                except BaseException as e:
                    completed.append((ev, f, t0, e))
            if not completed:
                time.sleep(0.00005)
                continue
            for ev, f, t0, err in completed:
                with self._lock:
                    self._inflight_gpu.pop(ev, None)
                if err is not None:
                    self._results.put(FetchResult(f, None, error=err))
                else:
                    data = self._arena_buf[f.arena_off:f.arena_off + f.length]
                    self._results.put(FetchResult(
                        f, data,
                        latency_ms=(time.perf_counter() - t0) * 1e3))

    def _do_fetch_blocking(self, f: CoalescedFetch) -> None:
        """Blocking lane on the thread pool: host-segment preads, the
        cross-host TCP path, and uploads of host bytes into the arena."""
        from .gpu_plane import is_gpu_key
        t0 = time.perf_counter()
        arena_buf = self._arena_buf   # capture: stable across arena growth
        try:
            owner = split_key(f.key)[0]
            if f.arena_off >= 0:
                if not self.manager.is_remote_host(owner) and is_gpu_key(f.key):
                    self.manager.gpu.read_device_into(
                        f.key, f.addr, f.length,
                        arena_buf.data_ptr() + f.arena_off)
                else:  # cross-host or host-spilled bytes: upload into place
                    import torch
                    raw = (self.manager.tcp_read(owner, f.key, f.addr, f.length)
                           if self.manager.is_remote_host(owner)
                           else self.manager.remote_read(f.key, f.addr, f.length))
                    if len(raw) != f.length:
                        raise FetchFailedError(
                            f"short read: {len(raw)}/{f.length} at key={f.key:#x}")
                    arena_buf[f.arena_off:f.arena_off + f.length] = \
                        torch.frombuffer(bytearray(raw), dtype=torch.uint8) \
                        .to(arena_buf.device)
                data = arena_buf[f.arena_off:f.arena_off + f.length]
            elif self.manager.is_remote_host(owner):
                data = self.manager.tcp_read(owner, f.key, f.addr, f.length)
            elif is_gpu_key(f.key):
                data = self.manager.remote_read_device(f.key, f.addr, f.length)
            else:
                data = self.manager.remote_read(f.key, f.addr, f.length)
            if len(data) != f.length:
                raise FetchFailedError(
                    f"short read: {len(data)}/{f.length} at key={f.key:#x}")
            self._results.put(FetchResult(
                f, data, latency_ms=(time.perf_counter() - t0) * 1e3))
        except BaseException as e:  # surfaced to consumer, fails the task
            self._results.put(FetchResult(f, None, error=e))
        finally:
            if f.arena_off >= 0:
                with self._lock:
                    self._blocking_arena_inflight -= 1

    # ------------------------------------------------------------------

    def __iter__(self) -> Iterator[tuple]:
        """Yields (BlockRef, memoryview) per block, in arrival order."""
        mgr = self.manager
        my_exec = mgr.executor_id
        try:
            while True:
                with self._lock:
                    if self._outstanding == 0 and self._hop2_pending == 0:
                        break
                t0 = time.perf_counter_ns()
                res = self._results.get()
                self.metrics.fetch_wait_ns += time.perf_counter_ns() - t0
                if res.error is not None:
                    with self._lock:
                        self._failed = res.error
                    raise FetchFailedError(
                        "fetch failed" + (
                            f" ({len(res.fetch.blocks)} blocks at "
                            f"key={res.fetch.key:#x})" if res.fetch else "")
                        + f": {res.error!r}") from res.error
                if res.fetch is None:      # sentinel: recheck termination
                    continue
                with self._lock:
                    self._bytes_in_flight -= res.fetch.length
                    src = split_key(res.fetch.key)[0]
                    self._reqs_by_source[src] = \
                        self._reqs_by_source.get(src, 1) - 1
                    self._outstanding -= 1
                self._pump()
                owner = split_key(res.fetch.key)[0]
                remote = owner != my_exec
                if remote:
                    self.metrics.remote_bytes_read += res.fetch.length
                    self.metrics.remote_blocks_fetched += len(res.fetch.blocks)
                    if mgr.reader_stats is not None:
                        mgr.reader_stats.update(owner, res.latency_ms)
                else:
                    self.metrics.local_bytes_read += res.fetch.length
                    self.metrics.local_blocks_fetched += len(res.fetch.blocks)
                data = res.data
                view = (memoryview(data)
                        if isinstance(data, (bytes, bytearray, memoryview))
                        else data)  # device tensor: torch slicing below
                off = 0
                for b in res.fetch.blocks:
                    yield b, view[off:off + b.length]
                    off += b.length
        finally:
            self._done.set()
            self._pool.shutdown(wait=False)
            lm = getattr(mgr, "lifetime_metrics", None)
            if lm is not None:   # test fakes may omit the rollup
                lm.merge(self.metrics)


class ShuffleReader:
    """Public reader: iterate raw blocks, or consume via the GENERIC
    aggregation/ordering hookup — the role the reference's reader plays
    once (deserialize -> aggregate -> ExternalSorter ordering,
    RdmaShuffleReader.scala:61-114) instead of every workload
    re-implementing it."""

    #: named reductions for read_aos(aggregator=...)
    AGGREGATORS = ("sum", "sum_f64", "count", "min", "max")

    def __init__(self, manager: ShuffleManager, handle: ShuffleHandle,
                 start_partition: int, end_partition: int, arena=None):
        self.manager = manager
        self.handle = handle
        self.start_partition = start_partition
        self.end_partition = end_partition
        self.fetcher = FetcherIterator(manager, handle,
                                       start_partition, end_partition,
                                       arena=arena)

    def __iter__(self):
        return iter(self.fetcher)

    def collect_partitions(self) -> dict:
        """partition -> list of data views (one per map task, unordered)."""
        out = {p: [] for p in range(self.start_partition, self.end_partition + 1)}
        for ref, data in self.fetcher:
            out[ref.partition].append(data)
        return out

    # ---------------- generic consumption (fixed-width AoS records) ----

    def _drain_aos_pairs(self):
        """CPU lane: fetch everything, return (keys u64, values u64)
        numpy arrays (the GPU lane drains into the fetch arena instead)."""
        from .writer import unpack_partition_segment
        import numpy as np
        ks, vs = [], []
        for _ref, data in self.fetcher:
            k, v = unpack_partition_segment(data, 8)
            ks.append(np.array(k))
            vs.append(np.ascontiguousarray(v).reshape(-1, 8)
                      .view(np.uint64).reshape(-1))
        if not ks:
            return None, None
        return np.concatenate(ks), np.concatenate(vs)

    def read_aos(self, aggregator: Optional[str] = None,
                 ordering: bool = False, end_bit: int = 64,
                 sort_tmp=None, sort_ws=None):
        """Consume the whole partition range of AoS (u64 key, u64 value)
        records, applying an optional keyed reduction and/or key ordering
        ON DEVICE. Returns (keys, values):

        * aggregator=None, ordering=True  -> records sorted by key bits
          [0, end_bit) (callers whose partitions share top bits pass a
          smaller end_bit and save radix passes)
        * aggregator in AGGREGATORS -> one row per distinct key, keys
          ascending; "sum_f64" treats the value payload as float64
        * both None/False -> unordered concatenated records
        """
        if aggregator is not None and aggregator not in self.AGGREGATORS:
            raise ValueError(f"unknown aggregator {aggregator!r}; "
                             f"expected one of {self.AGGREGATORS}")
        mgr = self.manager
        if mgr.gpu is not None:
            import torch
            for _ in self.fetcher:
                pass
            arena = self.fetcher.arena
            pairs = arena.view(torch.int64)
            if pairs.numel() == 0:
                empty = torch.empty(0, dtype=torch.int64,
                                    device=f"cuda:{mgr.gpu.device}")
                return empty, empty.clone()
            if aggregator is not None or ordering:
                from .ops.radix import sort_pairs_aos
                pairs = sort_pairs_aos(pairs, 0, end_bit, tmp=sort_tmp,
                                       ws=sort_ws)
            k = pairs[0::2]
            v = pairs[1::2]
            if aggregator is None:
                return k, v
            k = k.contiguous()
            v = v.contiguous()
            uk, inverse, cnt = torch.unique_consecutive(
                k, return_inverse=True, return_counts=True)
            if aggregator == "count":
                return uk, cnt
            if aggregator == "sum":
                out = torch.zeros(uk.numel(), dtype=torch.int64,
                                  device=k.device)
                out.index_add_(0, inverse, v)
                return uk, out
            if aggregator == "sum_f64":
                out = torch.zeros(uk.numel(), dtype=torch.float64,
                                  device=k.device)
                out.index_add_(0, inverse, v.view(torch.float64))
                return uk, out
            red = "amin" if aggregator == "min" else "amax"
            out = torch.empty(uk.numel(), dtype=torch.int64, device=k.device)
            out.scatter_reduce_(0, inverse, v, reduce=red,
                                include_self=False)
            return uk, out
        # ---- CPU path: numpy mirror of the device semantics ----
        import numpy as np
        k, v = self._drain_aos_pairs()
        if k is None:
            return np.array([], dtype=np.uint64), np.array([], dtype=np.uint64)
        if aggregator is not None or ordering:
            mask = np.uint64((1 << end_bit) - 1) if end_bit < 64 \
                else np.uint64(2 ** 64 - 1)
            order = np.argsort(k & mask, kind="stable")
            k, v = k[order], v[order]
        if aggregator is None:
            return k, v
        uk, start = np.unique(k, return_index=True)
        if aggregator == "count":
            return uk, np.diff(np.append(start, len(k)))
        if aggregator == "sum":
            return uk, np.add.reduceat(v, start)
        if aggregator == "sum_f64":
            return uk, np.add.reduceat(v.view(np.float64), start)
        fn = np.minimum if aggregator == "min" else np.maximum
        return uk, fn.reduceat(v, start)

    def dense_sum(self, key_lo: int, span: int, dtype: str = "f64"):
        """Keyed sum into a DENSE [key_lo, key_lo+span) vector without a
        sort — the iterative-workload (PageRank) reduction: every fetched
        chunk is index-added as it arrives, overlapping with in-flight
        fetches. dtype 'f64' treats payloads as float64, 'i64' as int64."""
        mgr = self.manager
        if mgr.gpu is not None:
            import torch
            tdtype = torch.float64 if dtype == "f64" else torch.int64
            sums = torch.zeros(span, dtype=tdtype,
                               device=f"cuda:{mgr.gpu.device}")
            from .utils import as_device_i64
            for _ref, data in self.fetcher:
                t = as_device_i64(data)
                idx = t[0::2] - key_lo
                val = t[1::2].contiguous()
                sums.index_add_(0, idx, val.view(tdtype)
                                if dtype == "f64" else val)
            return sums
        import numpy as np
        from .writer import unpack_partition_segment
        ndtype = np.float64 if dtype == "f64" else np.int64
        sums = np.zeros(span, dtype=ndtype)
        for _ref, data in self.fetcher:
            k, v = unpack_partition_segment(data, 8)
            idx = (np.asarray(k) - key_lo).astype(np.int64)
            np.add.at(sums, idx, np.ascontiguousarray(v).reshape(-1, 8)
                      .view(ndtype).reshape(-1))
        return sums

    def read_records(self):
        """Pickled-record lane: yields (key, value) python objects —
        the deserialize role of RdmaShuffleReader.scala:61-77."""
        import io
        import pickle
        for _ref, data in self.fetcher:
            if not isinstance(data, (bytes, bytearray, memoryview)):
                # arena mode landed the bytes in the device arena: one
                # bulk D2H (bytes() on a CUDA tensor would sync per byte)
                data = data.cpu().numpy().tobytes()
            bio = io.BytesIO(bytes(data))
            end = len(bio.getvalue())
            while bio.tell() < end:
                yield pickle.load(bio)

    @property
    def metrics(self) -> TaskMetrics:
        return self.fetcher.metrics
