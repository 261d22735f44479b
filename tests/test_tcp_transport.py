"""Cross-host fallback lane: TCP data servers (transport=tcp forces every
non-self read through the socket path — the multi-node configuration)."""

import numpy as np
import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.driver import Driver
from sparkrdma_amd.manager import ShuffleManager
from sparkrdma_amd.partitioner import HashPartitioner
from sparkrdma_amd.writer import unpack_partition_segment


@pytest.fixture
def tcp_cluster(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), transport="tcp",
                       max_buffer_allocation_size=1 << 30)
    driver = Driver(conf)
    managers = [ShuffleManager(conf, executor_id=i, driver_port=driver.port)
                for i in range(2)]
    import time
    deadline = time.monotonic() + 5
    while time.monotonic() < deadline and any(
            len(m._members) < 2 for m in managers):
        time.sleep(0.01)
    yield managers
    for m in managers:
        m.stop()
    driver.stop()


def test_tcp_shuffle_roundtrip(tcp_cluster):
    managers = tcp_cluster
    R = 8
    part = HashPartitioner(R)
    handle = managers[0].register_shuffle(num_maps=2, num_partitions=R)
    all_keys = []
    for mid, mgr in enumerate(managers):
        rng = np.random.default_rng(mid)
        keys = rng.integers(0, 2 ** 63, 20_000, dtype=np.uint64)
        all_keys.append(keys)
        w = mgr.get_writer(handle, mid)
        w.write_batch(keys, keys.view(np.uint8).reshape(-1, 8).copy())
        w.stop(True, partitioner=part)
    got = []
    for i, mgr in enumerate(managers):
        reader = mgr.get_reader(handle, i * 4, i * 4 + 3)
        for ref, data in reader:
            k, v = unpack_partition_segment(data, 8)
            assert np.array_equal(np.asarray(k).view(np.uint8).reshape(-1),
                                  np.asarray(v).reshape(-1))
            got.append(np.array(k))
        # every cross-executor byte went over TCP
        assert reader.metrics.remote_bytes_read > 0
    want = np.sort(np.concatenate(all_keys))
    assert np.array_equal(np.sort(np.concatenate(got)), want)


def test_tcp_server_rejects_foreign_key(tcp_cluster):
    m0, m1 = tcp_cluster
    from sparkrdma_amd.map_output import make_key
    # ask executor 1's server for executor 0's memory -> clean error
    with pytest.raises(IOError):
        m0._data_client.read("127.0.0.1", m1._data_server.port,
                             make_key(0, 2), 0, 64)
