"""Property-based invariants (hypothesis): pool, coalescer, RPC codec,
partitioners — the binary/bookkeeping contracts fuzzed."""

import numpy as np
from hypothesis import given, settings, strategies as st

from sparkrdma_amd.block_pool import MIN_BLOCK, BlockPool
from sparkrdma_amd.partitioner import HashPartitioner, RangePartitioner
from sparkrdma_amd.reader import BlockRef, coalesce_blocks
from sparkrdma_amd import rpc


@settings(max_examples=50, deadline=None)
@given(st.lists(st.integers(min_value=1, max_value=256 << 10), max_size=40))
def test_pool_alloc_free_invariants(sizes):
    slabs = {}
    nid = [2]

    def alloc(sz):
        sid = nid[0]
        nid[0] += 1
        slabs[sid] = sz
        return sid

    pool = BlockPool(1 << 20, 64 << 20, alloc, lambda s: slabs.pop(s))
    blocks = []
    for sz in sizes:
        b = pool.get(sz)
        assert b.capacity >= sz
        assert b.capacity >= MIN_BLOCK and (b.capacity & (b.capacity - 1)) == 0
        blocks.append(b)
    # no overlaps within a segment
    by_seg = {}
    for b in blocks:
        by_seg.setdefault(b.segment_id, []).append((b.offset, b.offset + b.capacity))
    for ranges in by_seg.values():
        ranges.sort()
        for (s1, e1), (s2, e2) in zip(ranges, ranges[1:]):
            assert e1 <= s2
    for b in blocks:
        b.release()
    assert pool.stats.used_bytes == 0


@settings(max_examples=50, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 3), st.integers(0, 1 << 20),
                          st.integers(0, 4096)), max_size=50),
       st.integers(1 << 12, 1 << 20))
def test_coalesce_preserves_blocks_and_bounds(specs, max_bytes):
    blocks = [BlockRef(0, i, key, addr, ln)
              for i, (key, addr, ln) in enumerate(specs)]
    out = coalesce_blocks(blocks, max_bytes)
    flat = [b for c in out for b in c.blocks]
    assert flat == [b for b in blocks if b.length > 0]  # order + completeness
    for c in out:
        assert c.length == sum(b.length for b in c.blocks)
        # contiguity within a fetch
        pos = c.addr
        for b in c.blocks:
            assert b.addr == pos and b.key == c.key
            pos += b.length
        assert c.length <= max(max_bytes, max(b.length for b in c.blocks))


@settings(max_examples=50, deadline=None)
@given(st.binary(max_size=30000), st.integers(0, 7),
       st.integers(64, 4096), st.integers(1, 1000))
def test_rpc_frame_roundtrip_any_chunking(body, mtype, max_frame, chunk):
    frames = rpc.encode_frames(mtype, body, max_frame)
    stream = b"".join(frames)
    dec = rpc.FrameDecoder()
    msgs = []
    for i in range(0, len(stream), chunk):
        msgs.extend(dec.feed(stream[i:i + chunk]))
    assert msgs == [(mtype, body)]


@settings(max_examples=30, deadline=None)
@given(st.integers(1, 64), st.lists(st.integers(0, 2 ** 64 - 1), min_size=1,
                                    max_size=500))
def test_partitioners_total_and_in_range(R, keys):
    k = np.array(keys, dtype=np.uint64)
    for part in (HashPartitioner(R), RangePartitioner.uniform(R)):
        pids = part.partition_ids(k)
        assert pids.min() >= 0 and pids.max() < R
        assert len(pids) == len(k)
    # range partitioner: pid is monotone in key
    rp = RangePartitioner.uniform(R)
    order = np.argsort(k)
    assert np.all(np.diff(rp.partition_ids(k[order])) >= 0)


@settings(max_examples=50, deadline=None)
@given(st.integers(0, 2**64 - 1), st.integers(-2**31, 2**31 - 1),
       st.integers(0, 2**32 - 1))
def test_block_location_pack_roundtrip(addr, length, key):
    from sparkrdma_amd.map_output import BlockLocation
    loc = BlockLocation(addr, length, key)
    assert BlockLocation.unpack(loc.pack()) == loc


@settings(max_examples=30, deadline=None)
@given(st.lists(st.tuples(st.integers(0, 2**63), st.integers(0, 2**31 - 1),
                          st.integers(1, 2**32 - 1)), min_size=1, max_size=64))
def test_map_output_table_roundtrip(entries):
    from sparkrdma_amd.map_output import BlockLocation, MapTaskOutput
    t = MapTaskOutput(len(entries))
    for i, (a, l, k) in enumerate(entries):
        t.put(i, a, l, k)
    parsed = MapTaskOutput.parse_locations(t.tobytes())
    assert parsed == [BlockLocation(a, l, k) for a, l, k in entries]
