"""Round-2 fetch-pipeline behaviors: per-source outstanding-read caps
(reference RdmaShuffleFetcherIterator.scala:82-83 — per channel there),
async hop-2 off the constructor's critical path (:297-311), and the
announce-time mesh pre-build (RdmaShuffleManager.scala:121-126)."""

import threading
import time

import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.manager import ShuffleHandle
from sparkrdma_amd.map_output import MapTaskOutput, make_key
from sparkrdma_amd.reader import FetcherIterator


class _MultiSourceManager:
    """Fake manager with NUM_SRC source executors; records per-source
    concurrent data reads to verify the per-source cap."""

    NUM_SRC = 4

    def __init__(self, conf, latency_s=0.004, maps_per_src=6):
        self.conf = conf
        self.executor_id = 0
        self.reader_stats = None
        self.gpu = None
        self.maps_per_src = maps_per_src
        self._lock = threading.Lock()
        self._latency = latency_s
        self.inflight = {s: 0 for s in range(1, self.NUM_SRC + 1)}
        self.max_inflight = {s: 0 for s in range(1, self.NUM_SRC + 1)}
        self.hop2_calls = []

    def get_map_task_output_table(self, handle):
        # maps round-robin over source executors 1..NUM_SRC; table key's
        # exec id IS the owner (meta segment id 1)
        out = []
        for m in range(handle.num_maps):
            src = 1 + m % self.NUM_SRC
            out.append((m * 1000, make_key(src, 1)))
        return out

    def is_remote_host(self, exec_id):
        return False

    def remote_read(self, key, addr, length):
        src = key >> 16
        seg = key & 0xFFFF
        if seg == 1:  # hop-2 table read
            self.hop2_calls.append((src, addr))
            span = length // 16
            t = MapTaskOutput(span)
            for i in range(span):
                # every block non-contiguous => one fetch per block
                t.put(i, addr * 7919 + i * 1000, 64, make_key(src, 2))
            return t.tobytes()
        with self._lock:
            self.inflight[src] += 1
            self.max_inflight[src] = max(self.max_inflight[src],
                                         self.inflight[src])
        time.sleep(self._latency)
        with self._lock:
            self.inflight[src] -= 1
        return b"y" * length


def _mk_handle(num_maps, parts):
    return ShuffleHandle(0, num_maps, parts, "/nonexistent")


def test_per_source_request_cap():
    conf = ShuffleConf(read_requests_limit=2, max_bytes_in_flight=1 << 30)
    mgr = _MultiSourceManager(conf)
    n_maps = mgr.NUM_SRC * mgr.maps_per_src
    it = FetcherIterator(mgr, _mk_handle(n_maps, 4), 0, 3,
                         num_workers=16, seed=3)
    blocks = list(it)
    assert len(blocks) == n_maps * 4
    for src, peak in mgr.max_inflight.items():
        assert peak <= 2, f"source {src} exceeded per-source cap: {peak}"


def test_sources_progress_concurrently():
    """With a per-source cap of 1 and real latency, different sources'
    fetches must still overlap (one hot source cannot starve the rest)."""
    conf = ShuffleConf(read_requests_limit=1, max_bytes_in_flight=1 << 30)
    mgr = _MultiSourceManager(conf, latency_s=0.01, maps_per_src=4)
    n_maps = mgr.NUM_SRC * mgr.maps_per_src
    t0 = time.perf_counter()
    it = FetcherIterator(mgr, _mk_handle(n_maps, 2), 0, 1,
                         num_workers=16, seed=5)
    blocks = list(it)
    dt = time.perf_counter() - t0
    assert len(blocks) == n_maps * 2
    total_fetches = n_maps * 2  # non-contiguous: no coalescing
    serial_time = total_fetches * 0.01
    # 4 sources at cap 1 => ~4x parallelism; allow generous slack
    assert dt < serial_time * 0.7, (dt, serial_time)


def test_constructor_not_blocked_by_hop2():
    """Hop 1+2 run async: constructing the iterator returns quickly even
    when metadata reads are slow and numMaps is large (VERDICT r01:
    'O(M) stall at Spark-scale map counts')."""
    conf = ShuffleConf()

    class SlowHop2(_MultiSourceManager):
        def remote_read(self, key, addr, length):
            if (key & 0xFFFF) == 1:
                time.sleep(0.005)
            return super().remote_read(key, addr, length)

    mgr = SlowHop2(conf, latency_s=0.0)
    n_maps = 128  # 32 hop-2 reads per source at 5 ms each
    t0 = time.perf_counter()
    it = FetcherIterator(mgr, _mk_handle(n_maps, 2), 0, 1, num_workers=8)
    construct_s = time.perf_counter() - t0
    assert construct_s < 0.1, construct_s
    blocks = list(it)
    assert len(blocks) == n_maps * 2


def test_hop2_grouped_per_source():
    conf = ShuffleConf()
    mgr = _MultiSourceManager(conf, latency_s=0.0)
    n_maps = mgr.NUM_SRC * mgr.maps_per_src
    it = FetcherIterator(mgr, _mk_handle(n_maps, 2), 0, 1, num_workers=4)
    list(it)
    # every map's table was read exactly once
    assert len(mgr.hop2_calls) == n_maps


def test_error_in_hop2_fails_task():
    from sparkrdma_amd.reader import FetchFailedError
    conf = ShuffleConf()

    class Failing(_MultiSourceManager):
        def remote_read(self, key, addr, length):
            if (key & 0xFFFF) == 1 and (key >> 16) == 2:
                raise OSError("metadata segment gone")
            return super().remote_read(key, addr, length)

    mgr = Failing(conf, latency_s=0.0)
    it = FetcherIterator(mgr, _mk_handle(8, 2), 0, 1, num_workers=4)
    with pytest.raises(FetchFailedError):
        list(it)


def test_prebuild_opens_peer_segments(tmp_path):
    """After announce, each executor's segment registry holds pre-opened
    readers for every peer's metadata segment — the host-plane analog of
    the reference's background channel pre-build."""
    from sparkrdma_amd.driver import Driver
    from sparkrdma_amd.manager import ShuffleManager

    conf = ShuffleConf(shm_dir=str(tmp_path),
                       max_buffer_allocation_size=1 << 30)
    driver = Driver(conf)
    conf.driver_port = driver.port
    managers = [ShuffleManager(conf, executor_id=i, driver_port=driver.port)
                for i in range(4)]
    try:
        deadline = time.monotonic() + 5
        want = {i: {make_key(j, 1) for j in range(4) if j != i}
                for i in range(4)}
        while time.monotonic() < deadline:
            if all(want[i] <= set(m._registry._readers)
                   for i, m in enumerate(managers)):
                break
            time.sleep(0.02)
        for i, m in enumerate(managers):
            assert want[i] <= set(m._registry._readers), \
                f"executor {i} did not pre-open peers' metadata segments"
    finally:
        for m in managers:
            m.stop()
        driver.stop()
