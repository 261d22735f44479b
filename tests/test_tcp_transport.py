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
            if not isinstance(data, (bytes, bytearray, memoryview)):
                data = data.cpu().numpy().tobytes()   # GPU-box arena mode
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


def test_tcp_client_pool_survives_rejections(tcp_cluster):
    """A protocol-level rejection (status<0) must release the pooled
    connection exactly once: repeated rejections past MAX_CONNS_PER_PEER
    must neither deadlock _acquire nor corrupt the pool counts, and the
    same client must still serve valid reads afterwards."""
    m0, m1 = tcp_cluster
    from sparkrdma_amd.data_server import DataClient
    from sparkrdma_amd.map_output import make_key
    client = m0._data_client
    ep_port = m1._data_server.port
    for _ in range(DataClient.MAX_CONNS_PER_PEER * 3):
        with pytest.raises(IOError):
            client.read("127.0.0.1", ep_port, make_key(0, 2), 0, 64)
    ep = ("127.0.0.1", ep_port)
    assert client._counts.get(ep, 0) == len(client._free.get(ep, [])), \
        "every rejected read must return its socket to the free list"
    assert 0 <= client._counts.get(ep, 0) <= DataClient.MAX_CONNS_PER_PEER


@pytest.fixture
def tcp_cluster_compressed(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), transport="tcp",
                       max_buffer_allocation_size=1 << 30,
                       tcp_compress=True, tcp_chunk_size=64 << 10)
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


def test_tcp_compressed_chunked_roundtrip(tcp_cluster_compressed):
    """Deflate codec + small chunks: multi-chunk responses reassemble
    bit-exact (the wrapStream analog on the one lane where a codec pays —
    VERDICT r01 item 8)."""
    managers = tcp_cluster_compressed
    R = 4
    part = HashPartitioner(R)
    handle = managers[0].register_shuffle(num_maps=2, num_partitions=R)
    all_keys = []
    for mid, mgr in enumerate(managers):
        rng = np.random.default_rng(100 + mid)
        # compressible payload: low-entropy values
        keys = rng.integers(0, 2 ** 63, 50_000, dtype=np.uint64)
        vals = np.zeros((50_000, 8), dtype=np.uint8)
        vals[:, 0] = (keys & np.uint64(0xFF)).astype(np.uint8)
        all_keys.append(keys)
        w = mgr.get_writer(handle, mid)
        w.write_batch(keys, vals)
        w.stop(True, partitioner=part)
    got = []
    for i, mgr in enumerate(managers):
        reader = mgr.get_reader(handle, i * 2, i * 2 + 1)
        for ref, data in reader:
            if not isinstance(data, (bytes, bytearray, memoryview)):
                data = data.cpu().numpy().tobytes()   # GPU-box arena mode
            k, v = unpack_partition_segment(data, 8)
            assert np.array_equal(
                np.asarray(v)[:, 0],
                (np.asarray(k) & np.uint64(0xFF)).astype(np.uint8))
            got.append(np.array(k))
        assert reader.metrics.remote_bytes_read > 0
    want = np.sort(np.concatenate(all_keys))
    assert np.array_equal(np.sort(np.concatenate(got)), want)


def test_tcp_chunked_large_single_read(tcp_cluster):
    """One read far bigger than the chunk size streams correctly."""
    m0, m1 = tcp_cluster
    from sparkrdma_amd.map_output import make_key
    blk = m1.pool.get(3_000_000)
    seg = m1.data_segment(blk.segment_id)
    rng = np.random.default_rng(9)
    payload = rng.integers(0, 256, 3_000_000, dtype=np.uint8).tobytes()
    seg.write(blk.offset, payload)
    m1.conf.tcp_chunk_size  # default 4M; shrink via server attr
    m1._data_server._chunk = 256 << 10
    got = m0._data_client.read("127.0.0.1", m1._data_server.port,
                               make_key(1, blk.segment_id), blk.offset,
                               len(payload))
    assert got == payload
    blk.release()


def test_tcp_client_pool_parallel_reads(tcp_cluster):
    """Concurrent fetches to ONE peer fan out over pooled connections."""
    import threading
    m0, m1 = tcp_cluster
    from sparkrdma_amd.map_output import make_key
    blk = m1.pool.get(1 << 20)
    seg = m1.data_segment(blk.segment_id)
    data = bytes(range(256)) * 4096
    seg.write(blk.offset, data)
    errs = []

    def worker():
        try:
            for _ in range(10):
                got = m0._data_client.read(
                    "127.0.0.1", m1._data_server.port,
                    make_key(1, blk.segment_id), blk.offset, len(data))
                assert got == data
        except Exception as e:  # pragma: no cover
            errs.append(e)

    ts = [threading.Thread(target=worker) for _ in range(8)]
    for t in ts:
        t.start()
    for t in ts:
        t.join()
    assert not errs
    assert m0._data_client._counts[("127.0.0.1", m1._data_server.port)] > 1
    blk.release()


def test_tcp_server_rejects_oversize_read(tcp_cluster):
    """length > MAX_READ gets a clean -1 status before any streaming
    starts, and the connection stays usable for the next request."""
    m0, m1 = tcp_cluster
    from sparkrdma_amd import data_server as ds
    from sparkrdma_amd.map_output import make_key
    with pytest.raises(IOError):
        m0._data_client.read("127.0.0.1", m1._data_server.port,
                             make_key(1, 2), 0, ds.MAX_READ + 1)
    # pool still consistent, later rejection also clean
    with pytest.raises(IOError):
        m0._data_client.read("127.0.0.1", m1._data_server.port,
                             make_key(0, 2), 0, 64)
