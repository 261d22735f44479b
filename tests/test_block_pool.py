"""Block-pool semantics (reference RdmaBufferManager.java:93-211: pow-2
size classes, 16 KiB min block, pooled reuse, idle trimming)."""

import pytest

from sparkrdma_amd.block_pool import MIN_BLOCK, Block, BlockPool, _round_pow2


def make_pool(slab_size=1 << 20, max_bytes=4 << 20):
    slabs = {}
    freed = []
    next_id = [2]

    def alloc(size):
        sid = next_id[0]
        next_id[0] += 1
        slabs[sid] = size
        return sid

    def free(sid):
        freed.append(sid)
        del slabs[sid]

    pool = BlockPool(slab_size, max_bytes, alloc, free)
    return pool, slabs, freed


def test_round_pow2():
    assert _round_pow2(1) == MIN_BLOCK
    assert _round_pow2(MIN_BLOCK) == MIN_BLOCK
    assert _round_pow2(MIN_BLOCK + 1) == MIN_BLOCK * 2
    assert _round_pow2(100 << 10) == 128 << 10


def test_basic_alloc_free_reuse():
    pool, slabs, _ = make_pool()
    b1 = pool.get(20 << 10)          # rounds to 32k
    assert b1.capacity == 32 << 10
    assert len(slabs) == 1
    off1 = (b1.segment_id, b1.offset)
    b1.release()
    b2 = pool.get(32 << 10)          # reuses the freed block
    assert (b2.segment_id, b2.offset) == off1
    b2.release()
    assert pool.stats.used_bytes == 0


def test_buddy_coalescing():
    pool, _, _ = make_pool(slab_size=256 << 10, max_bytes=256 << 10)
    blocks = [pool.get(MIN_BLOCK) for _ in range(16)]  # fills the slab
    assert pool.stats.used_bytes == 256 << 10
    with pytest.raises(MemoryError):
        pool.get(MIN_BLOCK)  # exhausted, cannot grow past max_bytes
    for b in blocks:
        b.release()
    # after coalescing a full-slab allocation must succeed
    big = pool.get(256 << 10)
    assert big.capacity == 256 << 10
    big.release()


def test_mixed_sizes_share_slab():
    pool, slabs, _ = make_pool(slab_size=1 << 20, max_bytes=1 << 20)
    a = pool.get(512 << 10)
    b = pool.get(256 << 10)
    c = pool.get(128 << 10)
    d = pool.get(128 << 10)
    assert len(slabs) == 1
    # distinct, non-overlapping ranges
    ranges = sorted((x.offset, x.offset + x.capacity) for x in (a, b, c, d))
    for (s1, e1), (s2, e2) in zip(ranges, ranges[1:]):
        assert e1 <= s2
    for x in (a, b, c, d):
        x.release()


def test_slab_growth_and_trim():
    pool, slabs, freed = make_pool(slab_size=1 << 20, max_bytes=4 << 20)
    blocks = [pool.get(1 << 20) for _ in range(4)]
    assert pool.stats.slab_count == 4
    # freeing everything pushes idle above 0.9*max -> trim to <= 0.65*max
    for b in blocks:
        b.release()
    assert pool.stats.slab_bytes <= int(0.65 * (4 << 20)) + (1 << 20)
    assert freed  # some slabs actually returned to the backend


def test_oversize_request_rejected():
    pool, _, _ = make_pool(slab_size=1 << 20)
    with pytest.raises(MemoryError):
        pool.get(2 << 20)
    with pytest.raises(ValueError):
        pool.get(0)


def test_refcounting():
    pool, _, _ = make_pool()
    b = pool.get(MIN_BLOCK)
    b.retain()
    b.release()
    assert pool.stats.frees == 0   # still referenced
    b.release()
    assert pool.stats.frees == 1


def test_preallocate_warms_pool():
    pool, slabs, _ = make_pool()
    pool.preallocate(64 << 10, 8)
    assert pool.stats.used_bytes == 0
    assert pool.stats.slab_count >= 1
    # subsequent gets hit the warmed slab, no growth
    before = pool.stats.slab_count
    blocks = [pool.get(64 << 10) for _ in range(8)]
    assert pool.stats.slab_count == before
    for b in blocks:
        b.release()


def test_buddy_fuzz_against_model():
    """Random alloc/free sequence vs an interval model: no live block may
    overlap another, every block stays inside its slab, and draining all
    blocks returns every slab to fully-free (perfect coalescing)."""
    import random
    rng = random.Random(9)
    SLAB = 1 << 22          # 4 MiB
    next_id = [0]

    def alloc_slab(size):
        next_id[0] += 1
        return next_id[0]

    pool = BlockPool(SLAB, 8 * SLAB, alloc_slab)
    live = []               # (block, (seg, lo, hi))
    for step in range(3000):
        if live and (rng.random() < 0.45 or len(live) > 400):
            i = rng.randrange(len(live))
            b, _ = live.pop(i)
            b.release()
        else:
            size = rng.choice((1, 100, 16 << 10, 40 << 10, 64 << 10,
                               1 << 20, (1 << 22) - 7))
            try:
                b = pool.get(size)
            except MemoryError:
                continue
            assert b.capacity >= size
            lo, hi = b.offset, b.offset + b.capacity
            assert 0 <= lo and hi <= SLAB
            for _ob, (seg, olo, ohi) in live:
                if seg == b.segment_id:
                    assert hi <= olo or lo >= ohi, "overlapping live blocks"
            live.append((b, (b.segment_id, lo, hi)))
    for b, _ in live:
        b.release()
    assert pool.stats.used_bytes == 0
    for slab in pool._slabs.values():
        assert slab.fully_free, "coalescing must restore full slabs"
