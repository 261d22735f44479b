"""End-to-end shuffle over the host one-sided data plane (no GPU).

The analog of SURVEY §4 config 1 (local-cluster groupByKey): multiple
executors (in-process here; multi-process in test_multiprocess.py) write a
shuffle, every executor reads its partitions one-sidedly, and the merged
result equals the input.
"""

import numpy as np
import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.driver import Driver
from sparkrdma_amd.manager import ShuffleManager
from sparkrdma_amd.partitioner import HashPartitioner
from sparkrdma_amd.reader import FetchFailedError
from sparkrdma_amd.segments import FIRST_DATA_SEGMENT_ID
from sparkrdma_amd.writer import unpack_partition_segment


@pytest.fixture
def cluster(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), max_buffer_allocation_size=1 << 30)
    driver = Driver(conf)
    conf.driver_port = driver.port
    managers = [ShuffleManager(conf, executor_id=i, driver_port=driver.port)
                for i in range(3)]
    yield driver, managers
    for m in managers:
        m.stop()
    driver.stop()


def test_membership_announce(cluster):
    driver, managers = cluster
    import time
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline:
        if all(len(m._members) == 3 for m in managers):
            break
        time.sleep(0.01)
    for m in managers:
        assert set(m._members) == {0, 1, 2}
        assert m.app_id == driver.app_id


def test_fixed_width_shuffle_roundtrip(cluster):
    _, managers = cluster
    R = 8
    VW = 8  # value width
    part = HashPartitioner(R)
    handle = managers[0].register_shuffle(num_maps=3, num_partitions=R)
    rng = np.random.default_rng(7)
    all_keys = []
    for map_id, mgr in enumerate(managers):
        keys = rng.integers(0, 2 ** 63, 5000, dtype=np.uint64)
        values = keys.view(np.uint8).reshape(-1, 8).copy()  # value = key bytes
        all_keys.append(keys)
        w = mgr.get_writer(handle, map_id)
        w.write_batch(keys, values)
        w.stop(True, partitioner=part)

    # each executor reads a disjoint partition range
    got_keys = []
    for i, mgr in enumerate(managers):
        lo = i * R // 3
        hi = (i + 1) * R // 3 - 1
        reader = mgr.get_reader(handle, lo, hi)
        parts = reader.collect_partitions()
        for p, chunks in parts.items():
            for chunk in chunks:
                k, v = unpack_partition_segment(chunk, VW)
                # value integrity: value bytes == key bytes
                assert np.array_equal(k.view(np.uint8), v.reshape(-1))
                # partition correctness
                assert np.all(part.partition_ids(np.asarray(k)) == p)
                got_keys.append(np.array(k))
        # metrics: remote + local bytes observed
        assert reader.metrics.remote_blocks_fetched + \
            reader.metrics.local_blocks_fetched > 0
    got = np.sort(np.concatenate(got_keys))
    want = np.sort(np.concatenate(all_keys))
    assert np.array_equal(got, want)
    managers[0].unregister_shuffle(handle.shuffle_id)


def test_bytes_record_shuffle_groupby(cluster):
    """groupByKey over pickled python records — the plumbing config."""
    import pickle
    _, managers = cluster
    R = 4
    handle = managers[0].register_shuffle(num_maps=3, num_partitions=R)
    n_per_map = 1000
    for map_id, mgr in enumerate(managers):
        w = mgr.get_writer(handle, map_id)
        records = [(f"k{i % 50}", (map_id, i)) for i in range(n_per_map)]
        w.write_records(records, None)
        w.stop(True)
    # single executor reads all partitions and groups
    reader = managers[0].get_reader(handle, 0, R - 1)
    groups = {}
    for ref, data in reader:
        buf = bytes(data)
        off = 0
        while off < len(buf):
            obj, off = _unpickle_one(buf, off)
            groups.setdefault(obj[0], []).append(obj[1])
    assert len(groups) == 50
    assert sum(len(v) for v in groups.values()) == 3 * n_per_map
    for k, vals in groups.items():
        assert len(vals) == 3 * n_per_map // 50


def _unpickle_one(buf, off):
    import pickle
    import io
    bio = io.BytesIO(buf[off:])
    obj = pickle.load(bio)
    return obj, off + bio.tell()


def test_empty_partitions(cluster):
    _, managers = cluster
    R = 64  # more partitions than records -> many empty
    part = HashPartitioner(R)
    handle = managers[0].register_shuffle(num_maps=1, num_partitions=R)
    keys = np.arange(10, dtype=np.uint64)
    w = managers[0].get_writer(handle, 0)
    w.write_batch(keys)
    w.stop(True, partitioner=part)
    reader = managers[1].get_reader(handle, 0, R - 1)
    total = []
    for ref, data in reader:
        k, _ = unpack_partition_segment(data, 0)
        total.append(np.array(k))
    assert np.array_equal(np.sort(np.concatenate(total)), keys)


def test_unpublished_map_times_out(cluster):
    _, managers = cluster
    conf_backup = managers[1].conf.partition_location_fetch_timeout_ms
    managers[1].conf.partition_location_fetch_timeout_ms = 200
    try:
        handle = managers[0].register_shuffle(num_maps=2, num_partitions=2)
        w = managers[0].get_writer(handle, 0)
        w.write_batch(np.arange(5, dtype=np.uint64))
        w.stop(True, partitioner=HashPartitioner(2))
        # map 1 never publishes; hop 1/2 run async off the constructor
        # (r02), so the timeout surfaces on iteration as a fetch failure
        reader = managers[1].get_reader(handle, 0, 1)
        with pytest.raises(FetchFailedError) as ei:
            list(reader)
        assert isinstance(ei.value.__cause__, TimeoutError)
    finally:
        managers[1].conf.partition_location_fetch_timeout_ms = conf_backup


def test_barrier(cluster):
    _, managers = cluster
    import threading
    results = []

    def go(m):
        m.barrier()
        results.append(m.executor_id)

    threads = [threading.Thread(target=go, args=(m,)) for m in managers]
    for t in threads:
        t.start()
    for t in threads:
        t.join(timeout=10)
    assert sorted(results) == [0, 1, 2]


def test_concurrent_shuffles(cluster):
    """Two shuffles in flight: registrations, writes and reads interleave
    without cross-talk (per-shuffle tables and block liveness)."""
    _, managers = cluster
    part = HashPartitioner(4)
    h1 = managers[0].register_shuffle(num_maps=3, num_partitions=4)
    h2 = managers[0].register_shuffle(num_maps=3, num_partitions=4)
    rng = np.random.default_rng(21)
    data = {h.shuffle_id: [] for h in (h1, h2)}
    for mid, mgr in enumerate(managers):
        for h, base in ((h1, 0), (h2, 1 << 32)):
            keys = rng.integers(base, base + (1 << 31), 3000, dtype=np.uint64)
            data[h.shuffle_id].append(keys)
            w = mgr.get_writer(h, mid)
            w.write_batch(keys)
            w.stop(True, partitioner=part)
    for h in (h1, h2):
        reader = managers[1].get_reader(h, 0, 3)
        got = []
        for ref, chunk in reader:
            k, _ = unpack_partition_segment(chunk, 0)
            got.append(np.array(k))
        want = np.sort(np.concatenate(data[h.shuffle_id]))
        assert np.array_equal(np.sort(np.concatenate(got)), want)
    managers[0].unregister_shuffle(h1.shuffle_id)
    # h2 must still be readable after h1 is gone
    reader = managers[2].get_reader(h2, 0, 3)
    total = sum(len(unpack_partition_segment(c, 0)[0]) for _, c in reader)
    assert total == 9000
    managers[0].unregister_shuffle(h2.shuffle_id)


def test_metadata_tables_recycle(cluster):
    """Table regions return to a free list at unregister — a long-running
    executor's metadata segment does not grow per shuffle (the reference
    returns table buffers to its pool, RdmaShuffleManager.scala:296)."""
    _, managers = cluster
    m = managers[0]
    part = HashPartitioner(4)

    def one_round():
        h = m.register_shuffle(num_maps=1, num_partitions=4)
        w = m.get_writer(h, 0)
        w.write_batch(np.arange(100, dtype=np.uint64))
        w.stop(True, partitioner=part)
        assert sum(1 for _ in m.get_reader(h, 0, 3)) >= 1
        m.unregister_shuffle(h.shuffle_id)

    one_round()
    bump_after_first = m._meta_bump
    for _ in range(10):
        one_round()
    assert m._meta_bump == bump_after_first, \
        "metadata tables were not recycled"


def test_host_segment_ids_recycle_with_revalidation(tmp_path):
    """Host slab ids RECYCLE (15-bit key space); a fetcher's cached fd
    revalidates the path's inode per read, so a reused id never serves
    the unlinked old file — the host-plane analog of the GPU slab
    generation check."""
    conf = ShuffleConf(shm_dir=str(tmp_path),
                       max_buffer_allocation_size=1 << 26)  # trims eagerly
    driver = Driver(conf)
    m0 = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    m1 = ShuffleManager(conf, executor_id=1, driver_port=driver.port)
    try:
        part = HashPartitioner(4)
        for round_ in range(8):
            h = m0.register_shuffle(num_maps=1, num_partitions=4)
            keys = np.arange(round_ * 1000, round_ * 1000 + 500,
                             dtype=np.uint64)
            w = m0.get_writer(h, 0)
            w.write_batch(keys, keys.view(np.uint8).reshape(-1, 8).copy())
            w.stop(True, partitioner=part)
            reader = m1.get_reader(h, 0, 3)
            got = []
            for _ref, data in reader:
                k, v = unpack_partition_segment(data, 8)
                assert np.array_equal(np.asarray(k).view(np.uint8),
                                      np.asarray(v).reshape(-1))
                got.append(np.array(k))
            assert np.array_equal(np.sort(np.concatenate(got)), keys), \
                f"round {round_}: stale host segment data"
            m0.unregister_shuffle(h.shuffle_id)
        # ids stayed bounded (recycled), not monotonically growing
        assert m0._next_segment_id <= FIRST_DATA_SEGMENT_ID + 4, \
            m0._next_segment_id
    finally:
        m0.stop()
        m1.stop()
        driver.stop()


def test_fd_count_bounded_over_many_shuffles(tmp_path):
    """A long-running executor's fd count must stay CONSTANT across many
    register/write/read/unregister cycles: segment-id recycling triggers
    reader revalidation, whose deferred-close list is bounded (the 45-min
    soak found 1 leaked fd per recycle before the bound existed)."""
    import os
    from sparkrdma_amd.engine import Engine

    import threading

    def nfds():
        return len(os.listdir("/proc/self/fd"))

    conf = ShuffleConf(shm_dir=str(tmp_path), max_buffer_allocation_size=1 << 30)
    with Engine(conf, rank=0, world_size=1, driver_port=0) as eng:
        baseline = None
        for i in range(120):
            h = eng.register_shuffle(1, 4)
            w = eng.manager.get_writer(h, 0)
            w.write_records([(k, k) for k in range(50)], None)
            w.stop(True)
            r = eng.manager.get_reader(h, 0, 3)
            assert sum(1 for _ in r.read_records()) == 50
            eng.unregister_shuffle(h)
            if i == 30:
                baseline = nfds()   # after caches/deferral window warm
                threads30 = threading.active_count()
        assert baseline is not None
        assert nfds() <= baseline + 4, \
            f"fd leak: {nfds()} vs baseline {baseline}"
        assert threading.active_count() <= threads30 + 2, \
            "thread leak across shuffle lifecycles"
