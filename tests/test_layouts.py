"""Binary layout contracts: 16-byte location entries, 12-byte driver
entries, key packing — the wire format the whole one-sided scheme rests on
(reference RdmaMapTaskOutput.scala:25-27)."""

import struct

import numpy as np
import pytest

from sparkrdma_amd.map_output import (BlockLocation, DriverTable, ENTRY_SIZE,
                                      MAP_ENTRY_SIZE, MapTaskOutput, make_key,
                                      split_key)


def test_entry_sizes():
    assert ENTRY_SIZE == 16
    assert MAP_ENTRY_SIZE == 12


def test_block_location_roundtrip():
    loc = BlockLocation(addr=0x1234_5678_9ABC, length=0x7FFF_0001, key=0xDEAD_BEEF)
    b = loc.pack()
    assert len(b) == 16
    assert BlockLocation.unpack(b) == loc
    # explicit layout: addr u64 LE, len i32 LE, key u32 LE
    addr, length, key = struct.unpack("<QiI", b)
    assert (addr, length, key) == (loc.addr, loc.length, loc.key)


def test_key_packing():
    k = make_key(executor_id=5, segment_id=3)
    assert split_key(k) == (5, 3)
    assert k == (5 << 16) | 3
    with pytest.raises(ValueError):
        make_key(1 << 16, 0)


def test_map_task_output_table():
    t = MapTaskOutput(8)
    assert t.nbytes == 128
    for p in range(8):
        t.put(p, addr=p * 1000, length=p * 10, key=make_key(1, 2))
    assert t.get(3) == BlockLocation(3000, 30, make_key(1, 2))
    # ranged raw read = what a remote fetcher sees
    raw = t.get_range(2, 4)
    locs = MapTaskOutput.parse_locations(raw)
    assert [l.addr for l in locs] == [2000, 3000, 4000]
    # full round trip
    t2 = MapTaskOutput.from_bytes(t.tobytes())
    assert t2.get(7) == t.get(7)


def test_map_task_output_shared_backing():
    buf = bytearray(4 * ENTRY_SIZE)
    t = MapTaskOutput(4, memoryview(buf))
    t.put(2, 42, 7, make_key(0, 1))
    # writes land in the backing buffer (what shm sharing relies on)
    addr, length, key = struct.unpack_from("<QiI", buf, 2 * ENTRY_SIZE)
    assert (addr, length, key) == (42, 7, make_key(0, 1))


def test_put_many_vectorized():
    t = MapTaskOutput(1000)
    addrs = np.arange(1000, dtype=np.uint64) * 16
    lens = np.full(1000, 16, dtype=np.int32)
    keys = np.full(1000, make_key(2, 2), dtype=np.uint32)
    t.put_many(addrs, lens, keys)
    assert t.get(999) == BlockLocation(999 * 16, 16, make_key(2, 2))


def test_driver_table():
    d = DriverTable(4)
    assert not d.complete
    for m in range(4):
        d.publish(m, addr=m * 64, key=make_key(m, 1))
    assert d.complete
    assert d.entry(2) == (128, make_key(2, 1))
    parsed = DriverTable.parse(d.tobytes())
    assert parsed[3] == (192, make_key(3, 1))
    with pytest.raises(IndexError):
        d.publish(4, 0, 1)


def test_partition_segment_codec_roundtrip():
    """pack/unpack_partition_segment: keys-only and keys+values, incl.
    empty — the CPU fixed-width wire format readers rely on."""
    import numpy as np
    from sparkrdma_amd.writer import pack_partition_segment, \
        unpack_partition_segment

    rng = np.random.default_rng(3)
    # keys only
    keys = rng.integers(0, 2 ** 64, 257, dtype=np.uint64)
    buf = pack_partition_segment(keys, None)
    k, v = unpack_partition_segment(buf, 0)
    assert np.array_equal(np.asarray(k), keys)
    assert v is None or len(v) == 0
    # keys + 12-byte values
    vals = rng.integers(0, 256, (257, 12), dtype=np.uint8)
    buf = pack_partition_segment(keys, vals)
    k, v = unpack_partition_segment(buf, 12)
    assert np.array_equal(np.asarray(k), keys)
    assert np.array_equal(np.asarray(v).reshape(-1, 12), vals)
    # empty
    empty = np.empty(0, dtype=np.uint64)
    buf = pack_partition_segment(empty, None)
    k, v = unpack_partition_segment(buf, 0)
    assert len(k) == 0
