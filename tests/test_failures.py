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
        import sparkrdma_amd.segments as seg_mod
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
