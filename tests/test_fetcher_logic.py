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
