"""Control-plane additions of round 2: the cross-host driver-table RPC
lane (hop 1 / publish when the driver's /dev/shm is unreachable), the
LOOKUP verification used by Engine.register_shuffle on non-zero ranks,
and the serialized driver-RPC correlation fix."""

import threading

import numpy as np
import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.driver import Driver
from sparkrdma_amd.manager import ShuffleHandle, ShuffleManager


@pytest.fixture
def pair(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), max_buffer_allocation_size=1 << 30)
    driver = Driver(conf)
    conf.driver_port = driver.port
    mgr = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    yield driver, mgr
    mgr.stop()
    driver.stop()


def test_lookup_shuffle(pair):
    driver, mgr = pair
    h = mgr.register_shuffle(num_maps=2, num_partitions=8)
    nm, np_, uri = mgr.lookup_shuffle(h.shuffle_id)
    assert (nm, np_) == (2, 8)
    assert uri == h.driver_table_path
    with pytest.raises(KeyError):
        mgr.lookup_shuffle(h.shuffle_id + 17)


def test_table_rpc_lane(pair):
    """A handle whose table path does not exist locally (cross-host driver)
    must route publish + hop-1 reads through the driver RPC lane and still
    converge (ADVICE r01 medium finding)."""
    driver, mgr = pair
    h = mgr.register_shuffle(num_maps=2, num_partitions=4)
    # simulate a remote driver: path that cannot be opened on this host
    remote_h = ShuffleHandle(h.shuffle_id, h.num_maps, h.num_partitions,
                             "/nonexistent/dir/driver_table")
    # publish both map outputs through the RPC lane
    for map_id in range(2):
        table, addr = mgr.alloc_table(h.num_partitions)
        for p in range(h.num_partitions):
            table.put(p, 0, 0, 0x10001)
        mgr.publish_map_output(remote_h, map_id, addr)
    entries = mgr.get_map_task_output_table(remote_h)
    assert len(entries) == 2
    assert all(key != 0 for _addr, key in entries)
    # the local-mmap view of the same shuffle agrees bit-for-bit
    mgr._cached_tables.clear()
    local_entries = mgr.get_map_task_output_table(h)
    assert local_entries == entries


def test_rpc_concurrent_correlation(pair):
    """Two threads issuing driver RPCs concurrently must each get their
    own reply (the r01 shared-FIFO race is closed by serializing request/
    response pairs)."""
    driver, mgr = pair
    handles = [mgr.register_shuffle(num_maps=1, num_partitions=1 + i)
               for i in range(4)]
    errors = []

    def worker(h):
        try:
            for _ in range(50):
                nm, np_, _ = mgr.lookup_shuffle(h.shuffle_id)
                assert (nm, np_) == (h.num_maps, h.num_partitions)
        except Exception as e:  # pragma: no cover
            errors.append(e)

    threads = [threading.Thread(target=worker, args=(h,)) for h in handles]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    assert not errors


def test_write_records_honors_partitioner(pair):
    _driver, mgr = pair
    h = mgr.register_shuffle(num_maps=1, num_partitions=4)
    w = mgr.get_writer(h, 0)
    # callable partitioner: everything to partition 2
    w.write_records([(k, k * 10) for k in range(20)],
                    partitioner=lambda k: 2)
    w.stop(True)
    reader = mgr.get_reader(h, 0, 3)
    parts = reader.collect_partitions()
    import pickle
    got = {p: [] for p in parts}
    for p, chunks in parts.items():
        for chunk in chunks:
            buf = bytes(chunk)
            off = 0
            while off < len(buf):
                obj = pickle.loads(buf[off:])
                frame = len(pickle.dumps(obj, protocol=4))
                got[p].append(obj)
                off += frame
    assert len(got[2]) == 20
    assert all(not got[p] for p in (0, 1, 3))


def test_write_records_partition_ids_interface(pair):
    _driver, mgr = pair
    from sparkrdma_amd.partitioner import RangePartitioner
    h = mgr.register_shuffle(num_maps=1, num_partitions=4)
    w = mgr.get_writer(h, 0)
    part = RangePartitioner.uniform(4)
    keys = np.array([0, 2 ** 62, 2 ** 63, 2 ** 63 + 2 ** 62], dtype=np.uint64)
    w.write_records([(int(k), 0) for k in keys], partitioner=part)
    w.stop(True)
    reader = mgr.get_reader(h, 0, 3)
    parts = reader.collect_partitions()
    sizes = {p: sum(len(c) for c in chunks) for p, chunks in parts.items()}
    assert all(sizes[p] > 0 for p in range(4))  # one record per quarter


def test_table_write_bounds_checked(pair):
    """A publish with an out-of-range map_id is rejected cleanly (ERROR
    reply), never corrupting adjacent metadata."""
    driver, mgr = pair
    h = mgr.register_shuffle(num_maps=2, num_partitions=4)
    from sparkrdma_amd import rpc
    mtype, _ = mgr._rpc_call(
        rpc.MSG_TABLE_WRITE, rpc.pack_table_write(h.shuffle_id, 99, 0, 1))
    assert mtype == rpc.MSG_ERROR


def test_late_joining_executor_announced_to_all(pair, tmp_path):
    """An executor that joins after the app is running must appear in
    every member's view (the announce fan-out re-sends full membership),
    and the newcomer sees the existing members too."""
    import time
    driver, mgr = pair
    late = ShuffleManager(ShuffleConf(shm_dir=mgr.conf.shm_dir,
                                      max_buffer_allocation_size=1 << 30),
                          executor_id=7, driver_port=driver.port)
    try:
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and (
                len(mgr._members) < 2 or len(late._members) < 2):
            time.sleep(0.01)
        assert set(mgr._members) == {0, 7}
        assert set(late._members) == {0, 7}
    finally:
        late.stop()


def test_double_stop_idempotent(tmp_path):
    """stop() twice on every component is a no-op, not an error — teardown
    runs from finally blocks and signal handlers, so it must be safe to
    repeat (and Engine.__exit__ may race an explicit shutdown())."""
    from sparkrdma_amd.engine import Engine
    conf = ShuffleConf(shm_dir=str(tmp_path), max_buffer_allocation_size=1 << 30)
    eng = Engine(conf, rank=0, world_size=1, driver_port=0)
    h = eng.register_shuffle(1, 2)
    w = eng.manager.get_writer(h, 0)
    w.write_records([(1, 2)], None)
    w.stop(True)
    eng.shutdown()
    eng.shutdown()           # second shutdown: no-op
    eng.manager.stop()       # direct repeats too
    eng.driver.stop()
