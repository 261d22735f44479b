"""Generic reader aggregation/ordering hookup (the shared
deserialize -> aggregate -> ordering role of RdmaShuffleReader.scala:
61-114) — CPU-lane semantics; the GPU lane shares the code path and is
covered by the GPU workload tests."""

import numpy as np
import pytest

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.driver import Driver
from sparkrdma_amd.manager import ShuffleManager
from sparkrdma_amd.partitioner import HashPartitioner


@pytest.fixture
def cluster(tmp_path):
    conf = ShuffleConf(shm_dir=str(tmp_path), max_buffer_allocation_size=1 << 30)
    driver = Driver(conf)
    conf.driver_port = driver.port
    managers = [ShuffleManager(conf, executor_id=i, driver_port=driver.port)
                for i in range(2)]
    yield managers
    for m in managers:
        m.stop()
    driver.stop()


def _write(managers, R, keys_per_map, vals_per_map, part):
    handle = managers[0].register_shuffle(len(managers), R)
    for mid, (mgr, k, v) in enumerate(zip(managers, keys_per_map,
                                          vals_per_map)):
        w = mgr.get_writer(handle, mid)
        w.write_batch(k, v.view(np.uint8).reshape(-1, 8).copy())
        w.stop(True, partitioner=part)
    return handle


def test_read_aos_sum(cluster):
    R = 4
    part = HashPartitioner(R)
    rng = np.random.default_rng(0)
    ks = [rng.integers(0, 50, 1000, dtype=np.uint64) for _ in range(2)]
    vs = [rng.integers(0, 1000, 1000, dtype=np.uint64) for _ in range(2)]
    handle = _write(cluster, R, ks, vs, part)
    reader = cluster[0].get_reader(handle, 0, R - 1)
    uk, sums = reader.read_aos(aggregator="sum")
    want = {}
    for k, v in zip(np.concatenate(ks), np.concatenate(vs)):
        want[int(k)] = want.get(int(k), 0) + int(v)
    got = dict(zip(uk.tolist(), sums.tolist()))
    assert got == want


@pytest.mark.parametrize("agg,npfn", [("min", np.minimum), ("max", np.maximum)])
def test_read_aos_minmax(cluster, agg, npfn):
    R = 4
    part = HashPartitioner(R)
    rng = np.random.default_rng(1)
    ks = [rng.integers(0, 20, 500, dtype=np.uint64) for _ in range(2)]
    vs = [rng.integers(0, 10**6, 500, dtype=np.uint64) for _ in range(2)]
    handle = _write(cluster, R, ks, vs, part)
    uk, out = cluster[0].get_reader(handle, 0, R - 1).read_aos(aggregator=agg)
    allk, allv = np.concatenate(ks), np.concatenate(vs)
    want = {}
    for k, v in zip(allk, allv):
        k = int(k)
        want[k] = int(v) if k not in want else int(npfn(want[k], int(v)))
    assert dict(zip(uk.tolist(), out.tolist())) == want


def test_read_aos_count_and_ordering(cluster):
    R = 4
    part = HashPartitioner(R)
    rng = np.random.default_rng(2)
    ks = [rng.integers(0, 30, 400, dtype=np.uint64) for _ in range(2)]
    vs = [np.arange(400, dtype=np.uint64) for _ in range(2)]
    handle = _write(cluster, R, ks, vs, part)
    uk, cnt = cluster[0].get_reader(handle, 0, R - 1).read_aos(
        aggregator="count")
    want = np.bincount(np.concatenate(ks).astype(np.int64))
    want_k = np.nonzero(want)[0]
    assert np.array_equal(uk.astype(np.int64), want_k)
    assert np.array_equal(cnt, want[want_k])
    # ordering only: full sorted stream, values ride along
    k2, v2 = cluster[0].get_reader(handle, 0, R - 1).read_aos(ordering=True)
    assert np.all(k2[1:] >= k2[:-1])
    assert len(k2) == 800


def test_read_aos_sum_f64(cluster):
    R = 4
    part = HashPartitioner(R)
    rng = np.random.default_rng(3)
    ks = [rng.integers(0, 10, 300, dtype=np.uint64) for _ in range(2)]
    fvals = [rng.random(300) for _ in range(2)]
    vs = [f.view(np.uint64) for f in fvals]
    handle = _write(cluster, R, ks, vs, part)
    uk, sums = cluster[0].get_reader(handle, 0, R - 1).read_aos(
        aggregator="sum_f64")
    want = {}
    for k, f in zip(np.concatenate(ks), np.concatenate(fvals)):
        want[int(k)] = want.get(int(k), 0.0) + float(f)
    for k, s in zip(uk.tolist(), sums.tolist()):
        assert abs(s - want[k]) < 1e-9


def test_dense_sum(cluster):
    R = 4
    part = HashPartitioner(R)   # hash part but dense_sum keys must cover
    # use keys within [0, 64) and a RangePartitioner-free dense target
    rng = np.random.default_rng(4)
    ks = [rng.integers(0, 64, 500, dtype=np.uint64) for _ in range(2)]
    fvals = [rng.random(500) for _ in range(2)]
    vs = [f.view(np.uint64) for f in fvals]
    handle = _write(cluster, R, ks, vs, part)
    sums = cluster[0].get_reader(handle, 0, R - 1).dense_sum(0, 64, "f64")
    want = np.zeros(64)
    np.add.at(want, np.concatenate(ks).astype(np.int64),
              np.concatenate(fvals))
    assert np.allclose(sums, want)


def test_read_records(cluster):
    R = 4
    handle = cluster[0].register_shuffle(2, R)
    for mid, mgr in enumerate(cluster):
        w = mgr.get_writer(handle, mid)
        w.write_records([(f"k{i % 5}", (mid, i)) for i in range(50)])
        w.stop(True)
    got = list(cluster[0].get_reader(handle, 0, R - 1).read_records())
    assert len(got) == 100
    keys = {k for k, _v in got}
    assert keys == {f"k{i}" for i in range(5)}


def test_read_aos_invalid_aggregator(cluster):
    handle = cluster[0].register_shuffle(2, 4)
    with pytest.raises(ValueError):
        cluster[0].get_reader(handle, 0, 3).read_aos(aggregator="median")
