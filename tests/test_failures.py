"""Failure semantics parity: executor-loss pruning +
re-announce, fetch failures surfacing as FetchFailedError, connect retry."""

import time

import numpy as np
import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.driver import Driver
from sparkrdma_amd.manager import ShuffleManager
from sparkrdma_amd.partitioner import HashPartitioner
from sparkrdma_amd.reader import FetchFailedError


def test_executor_loss_prunes_membership(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path))
    driver = Driver(conf)
    m0 = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    m1 = ShuffleManager(conf, executor_id=1, driver_port=driver.port)
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline and len(m0._members) < 2:
        time.sleep(0.01)
    assert set(m0._members) == {0, 1}
    m1.stop()  # clean departure closes the connection
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline and len(m0._members) != 1:
        time.sleep(0.01)
    assert set(m0._members) == {0}, "driver must re-announce without lost exec"
    m0.stop()
    driver.stop()


def test_fetch_failure_surfaces(tmp_path):
    """A dead serving executor (segments unlinked) must fail the read with
    FetchFailedError — Spark's stage-retry contract
    (RdmaShuffleFetcherIterator.scala:167)."""
    conf = ShuffleConf(shm_dir=str(tmp_path))
    driver = Driver(conf)
    m0 = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    m1 = ShuffleManager(conf, executor_id=1, driver_port=driver.port)
    try:
        handle = m0.register_shuffle(num_maps=2, num_partitions=4)
        part = HashPartitioner(4)
        for mid, mgr in enumerate((m0, m1)):
            w = mgr.get_writer(handle, mid)
            w.write_batch(np.arange(1000, dtype=np.uint64))
            w.stop(True, partitioner=part)
        # simulate executor-1 crash: its data segments vanish
        for seg in list(m1._data_segments.values()):
            seg.unlink()
        with pytest.raises((FetchFailedError, FileNotFoundError)):
            reader = m0.get_reader(handle, 0, 3)
            list(reader)
    finally:
        m0.stop()
        m1.stop()
        driver.stop()


def test_connect_retry_exhaustion(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), max_connection_attempts=2,
                       rdma_cm_event_timeout_ms=500)
    with pytest.raises(ConnectionError):
        ShuffleManager(conf, executor_id=0, driver_port=1)  # nothing listens


def test_read_after_unregister_fails_cleanly(tmp_path):
    """A reader constructed after unregister_shuffle must surface a fetch
    failure (the driver table is gone on both the mmap and RPC lanes),
    never hang or return stale data — the reference's liveness contract
    (blocks live only until unregisterShuffle,
    RdmaShuffleManager.scala:293-299)."""
    import numpy as np
    from sparkrdma_amd.conf import ShuffleConf
    from sparkrdma_amd.driver import Driver
    from sparkrdma_amd.manager import ShuffleManager
    from sparkrdma_amd.partitioner import HashPartitioner
    from sparkrdma_amd.reader import FetchFailedError

    conf = ShuffleConf(shm_dir=str(tmp_path),
                       partition_location_fetch_timeout_ms=500)
    driver = Driver(conf)
    mgr = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    try:
        part = HashPartitioner(4)
        handle = mgr.register_shuffle(num_maps=1, num_partitions=4)
        w = mgr.get_writer(handle, 0)
        w.write_batch(np.arange(100, dtype=np.uint64))
        w.stop(True, partitioner=part)
        mgr.unregister_shuffle(handle.shuffle_id)
        reader = mgr.get_reader(handle, 0, 3)
        with pytest.raises(FetchFailedError):
            list(reader)
    finally:
        mgr.stop()
        driver.stop()


def test_driver_death_fails_rpcs_fast(tmp_path):
    """Driver crash mid-run: manager RPCs must raise ConnectionError
    promptly (seconds, not the 30 s RPC timeout) once the connection
    drops — the analog of the reference failing tasks when the CM
    connection dies."""
    conf = ShuffleConf(shm_dir=str(tmp_path))
    driver = Driver(conf)
    mgr = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    try:
        h = mgr.register_shuffle(num_maps=1, num_partitions=2)
        driver.stop()
        t0 = time.monotonic()
        with pytest.raises((ConnectionError, TimeoutError)) as exc_info:
            for _ in range(10):  # first call may race the EOF notice
                mgr.lookup_shuffle(h.shuffle_id)
                time.sleep(0.05)
        assert time.monotonic() - t0 < 10.0, "must fail fast, not burn timeouts"
        assert isinstance(exc_info.value, ConnectionError)
        # and every later call fails instantly at entry
        t0 = time.monotonic()
        with pytest.raises(ConnectionError):
            mgr.lookup_shuffle(h.shuffle_id)
        assert time.monotonic() - t0 < 1.0
    finally:
        mgr.stop()
        driver.stop()


def test_barrier_released_on_member_loss(tmp_path):
    """If an executor dies while the others wait at the stage barrier, the
    driver must re-evaluate the barrier against the pruned membership and
    release the survivors — not leave them hanging."""
    conf = ShuffleConf(shm_dir=str(tmp_path))
    driver = Driver(conf)
    m0 = ShuffleManager(conf, executor_id=0, driver_port=driver.port)
    m1 = ShuffleManager(conf, executor_id=1, driver_port=driver.port)
    m2 = ShuffleManager(conf, executor_id=2, driver_port=driver.port)
    import threading
    try:
        deadline = time.monotonic() + 5
        while time.monotonic() < deadline and len(m0._members) < 3:
            time.sleep(0.01)
        assert len(m0._members) == 3
        done = []
        errs = []

        def wait_barrier(m):
            try:
                m.barrier(timeout=15)
                done.append(m.executor_id)
            except Exception as e:  # pragma: no cover
                errs.append(e)

        t0 = threading.Thread(target=wait_barrier, args=(m0,))
        t1 = threading.Thread(target=wait_barrier, args=(m1,))
        t0.start()
        t1.start()
        time.sleep(0.3)   # both waiting at the barrier
        m2.stop()         # third member dies instead of arriving
        t0.join(timeout=10)
        t1.join(timeout=10)
        assert not errs
        assert sorted(done) == [0, 1], "survivors must be released"
    finally:
        m0.stop()
        m1.stop()
        driver.stop()
