"""Pure fetch-pipeline logic: coalescing + partitioners (reference
RdmaShuffleFetcherIterator.scala:240-263 coalescing semantics)."""

import numpy as np

from sparkrdma_amd.partitioner import HashPartitioner, RangePartitioner
from sparkrdma_amd.reader import BlockRef, coalesce_blocks


def mk(map_id, part, key, addr, length):
    return BlockRef(map_id, part, key, addr, length)


def test_coalesce_contiguous_same_key():
    blocks = [mk(0, 0, 1, 0, 100), mk(0, 1, 1, 100, 50), mk(0, 2, 1, 150, 50)]
    out = coalesce_blocks(blocks, max_bytes=1000)
    assert len(out) == 1
    assert (out[0].addr, out[0].length) == (0, 200)
    assert len(out[0].blocks) == 3


def test_coalesce_respects_max_bytes():
    blocks = [mk(0, i, 1, i * 100, 100) for i in range(5)]
    out = coalesce_blocks(blocks, max_bytes=250)
    assert [c.length for c in out] == [200, 200, 100]


def test_coalesce_breaks_on_key_or_gap():
    blocks = [mk(0, 0, 1, 0, 100), mk(0, 1, 2, 100, 100),  # key change
              mk(0, 2, 2, 300, 100)]                        # gap
    out = coalesce_blocks(blocks, max_bytes=1000)
    assert len(out) == 3


def test_coalesce_drops_empty_blocks():
    blocks = [mk(0, 0, 1, 0, 0), mk(0, 1, 1, 0, 100), mk(0, 2, 1, 100, 0)]
    out = coalesce_blocks(blocks, max_bytes=1000)
    assert len(out) == 1
    assert out[0].length == 100


def test_oversize_single_block_kept_whole():
    out = coalesce_blocks([mk(0, 0, 1, 0, 5000)], max_bytes=100)
    assert len(out) == 1 and out[0].length == 5000


def test_hash_partitioner_range_and_determinism():
    p = HashPartitioner(37)
    keys = np.random.default_rng(0).integers(0, 2 ** 63, 10000, dtype=np.uint64)
    pids = p.partition_ids(keys)
    assert pids.min() >= 0 and pids.max() < 37
    assert np.array_equal(pids, p.partition_ids(keys))
    # roughly balanced
    counts = np.bincount(pids, minlength=37)
    assert counts.min() > 10000 / 37 * 0.5


def test_range_partitioner_terasort_property():
    rp = RangePartitioner.uniform(16)
    keys = np.random.default_rng(1).integers(0, 2 ** 64, 50000, dtype=np.uint64)
    pids = rp.partition_ids(keys)
    # all keys in partition i are < all keys in partition i+1
    for i in range(15):
        a = keys[pids == i]
        b = keys[pids == i + 1]
        if len(a) and len(b):
            assert a.max() < b.min()
    # boundary membership: bounds[i] itself belongs to partition i+1? No:
    # searchsorted(side=right) puts key == bounds[i] into partition i+1
    assert rp.partition_ids(np.array([rp.bounds[0]], dtype=np.uint64))[0] == 1
    assert rp.partition_ids(np.array([rp.bounds[0] - 1], dtype=np.uint64))[0] == 0


def test_parse_cpu_list():
    from sparkrdma_amd.reader import parse_cpu_list
    assert parse_cpu_list("") == []
    assert parse_cpu_list("0-3,8,10-11") == [0, 1, 2, 3, 8, 10, 11]
    assert parse_cpu_list("5") == [5]


class _FakeManager:
    """Minimal manager stub driving FetcherIterator's flow control."""

    def __init__(self, conf, latency_s=0.0):
        self.conf = conf
        self.executor_id = 0
        self.reader_stats = None
        self.max_seen_in_flight = 0
        self._in_flight = 0
        import threading
        self._lock = threading.Lock()
        self._latency = latency_s
        self.reads = []

    def get_map_task_output_table(self, handle):
        return [(0, 99)] * handle.num_maps  # table key 99 (executor 0)

    def is_remote_host(self, exec_id):
        return False

    gpu = None  # no device plane -> no arena mode

    def remote_read(self, key, addr, length):
        import time
        if key == 99:  # hop-2 table read: fabricate location entries
            from sparkrdma_amd.map_output import MapTaskOutput, make_key
            span = length // 16
            t = MapTaskOutput(span)
            for i in range(span):
                t.put(i, addr * 1000 + i * 100, 100, make_key(1, 2))
            return t.tobytes()
        with self._lock:
            self._in_flight += length
            self.max_seen_in_flight = max(self.max_seen_in_flight,
                                          self._in_flight)
        if self._latency:
            time.sleep(self._latency)
        self.reads.append((key, addr, length))
        with self._lock:
            self._in_flight -= length
        return b"x" * length


def _mk_handle(num_maps, parts):
    from sparkrdma_amd.manager import ShuffleHandle
    return ShuffleHandle(0, num_maps, parts, "/nonexistent")


def test_fetcher_respects_byte_budget():
    from sparkrdma_amd.conf import ShuffleConf
    from sparkrdma_amd.reader import FetcherIterator
    conf = ShuffleConf(max_bytes_in_flight=300 << 10,
                       shuffle_read_block_size=256 << 10)
    mgr = _FakeManager(conf, latency_s=0.002)
    it = FetcherIterator(mgr, _mk_handle(num_maps=16, parts=8), 0, 7,
                         num_workers=8, seed=1)
    blocks = list(it)
    assert len(blocks) == 16 * 8
    assert all(bytes(d) == b"x" * 100 for _, d in blocks)
    # in-flight bytes never exceeded the budget (one oversize block may
    # ride alone, but our blocks are 100B each)
    assert mgr.max_seen_in_flight <= conf.max_bytes_in_flight


def test_fetcher_randomization_deterministic_by_seed():
    from sparkrdma_amd.conf import ShuffleConf
    from sparkrdma_amd.reader import FetcherIterator
    conf = ShuffleConf()
    runs = []
    for _ in range(2):
        mgr = _FakeManager(conf)
        it = FetcherIterator(mgr, _mk_handle(4, 4), 0, 3,
                             num_workers=1, seed=7)
        list(it)
        runs.append([r[1] for r in mgr.reads])
    assert runs[0] == runs[1]


def test_range_partitioner_mulhi_equivalence():
    """The GPU kernel's func-2 partition (pid = floor(key * R / 2^64),
    multiply-high) must equal the CPU RangePartitioner's searchsorted
    over CEILING bounds for every key INCLUDING exact boundaries — the
    bit-for-bit contract the non-pow2 GPU path relies on."""
    rng = np.random.default_rng(77)
    for R in (2, 3, 7, 12, 100, 1000, 4096, 5000, 12000, 65521):
        p = RangePartitioner.uniform(R)
        keys = rng.integers(0, 2 ** 64, 5000, dtype=np.uint64)
        # inject exact boundary keys and their neighbours
        idx = rng.integers(0, R - 1, size=min(R - 1, 64))
        b = p.bounds[idx]
        keys = np.concatenate([keys, b, b - 1, b + 1,
                               np.array([0, 2**64 - 1], dtype=np.uint64)])
        want = p.partition_ids(keys)
        # mulhi in python ints (the kernel's (u128 k * R) >> 64)
        got = np.array([(int(k) * R) >> 64 for k in keys], dtype=np.int64)
        assert np.array_equal(got, want.astype(np.int64)), R


def test_hash_partitioner_mod_equivalence():
    """func-3 (splitmix mix then % R) equals HashPartitioner.partition_ids
    for non-pow2 R (the GPU hash-mod contract)."""
    rng = np.random.default_rng(78)
    MIX = 0xFF51AFD7ED558CCD
    for R in (3, 7, 100, 1000, 5000):
        p = HashPartitioner(R)
        keys = rng.integers(0, 2 ** 64, 3000, dtype=np.uint64)
        want = p.partition_ids(keys)
        got = []
        for k in keys:
            k = int(k)
            k ^= k >> 33
            k = (k * MIX) & ((1 << 64) - 1)
            k ^= k >> 33
            got.append(k % R)
        assert np.array_equal(np.array(got, dtype=np.int64),
                              want.astype(np.int64)), R
