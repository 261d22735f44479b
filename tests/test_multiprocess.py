"""Multi-process shuffle: real process isolation over the one-sided shm
plane — the CPU stand-in for the 8-executor xGMI topology. Also runs the
Engine wrapper the way torchrun does (RANK/WORLD_SIZE env)."""

import multiprocessing as mp
import os

import numpy as np
import pytest


def _worker(rank, world_size, driver_port, shm_dir, q):
    try:
        import sys
        sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        from sparkrdma_amd.conf import ShuffleConf
        from sparkrdma_amd.engine import Engine
        from sparkrdma_amd.partitioner import RangePartitioner
        from sparkrdma_amd.writer import unpack_partition_segment

        conf = ShuffleConf(shm_dir=shm_dir, max_buffer_allocation_size=1 << 30)
        eng = Engine(conf, rank=rank, world_size=world_size,
                     driver_port=driver_port)
        R = world_size * 2
        part = RangePartitioner.uniform(R, key_max=2 ** 32 - 1)
        # SPMD-collective registration: all ranks call, handles agree
        handle = eng.register_shuffle(num_maps=world_size, num_partitions=R)
        rng = np.random.default_rng(rank)
        keys = rng.integers(0, 2 ** 32, 20000, dtype=np.uint64)
        w = eng.manager.get_writer(handle, rank)
        w.write_batch(keys, keys.view(np.uint8).reshape(-1, 8).copy())
        w.stop(True, partitioner=part)
        eng.barrier()
        lo, hi = rank * 2, rank * 2 + 1
        reader = eng.manager.get_reader(handle, lo, hi)
        got = []
        for ref, data in reader:
            k, v = unpack_partition_segment(data, 8)
            assert np.array_equal(k.view(np.uint8), v.reshape(-1))
            assert lo <= ref.partition <= hi
            got.append(np.array(k))
        my_keys = np.sort(np.concatenate(got)) if got else np.array([], dtype=np.uint64)
        eng.barrier()
        q.put((rank, my_keys.tobytes(),
               reader.metrics.remote_bytes_read, reader.metrics.local_bytes_read))
        eng.barrier()
        eng.shutdown()
    except BaseException as e:  # surface failures to the parent
        import traceback
        q.put((rank, f"ERROR: {e}\n{traceback.format_exc()}", 0, 0))
        raise


def _pagerank_worker(rank, world, driver_port, shm_dir, q):
    try:
        import sys
        sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
        from sparkrdma_amd.conf import ShuffleConf
        from sparkrdma_amd.engine import Engine
        from sparkrdma_amd.workloads.pagerank import PageRank

        conf = ShuffleConf(shm_dir=shm_dir, max_buffer_allocation_size=1 << 30)
        eng = Engine(conf, rank=rank, world_size=world, driver_port=driver_port)
        pr = PageRank(eng, num_vertices=1 << 10, edges_per_executor=5000,
                      partitions_per_executor=8, device="cpu",
                      iterations=3, seed=11)
        pr.run_step()
        q.put((rank, pr.ranks.tobytes()))
        eng.barrier()
        eng.shutdown()
    except BaseException as e:
        import traceback
        q.put((rank, f"ERROR: {e}\n{traceback.format_exc()}"))
        raise


def test_multiprocess_pagerank_matches_dense(tmp_path):
    """2-process PageRank must equal the single-machine dense oracle —
    validates cross-process contribution routing + aggregation."""
    world = 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_pagerank_worker,
                         args=(r, world, port, str(tmp_path), q))
             for r in range(world)]
    for p in procs:
        p.start()
    parts = {}
    for _ in range(world):
        rank, payload = q.get(timeout=120)
        assert not isinstance(payload, str), f"rank {rank}: {payload}"
        parts[rank] = np.frombuffer(payload, dtype=np.float64)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    got = np.concatenate([parts[r] for r in range(world)])
    # rebuild the full edge list exactly as the workers did
    from sparkrdma_amd.workloads.pagerank import PageRank
    V = 1 << 10
    srcs, dsts = [], []
    for r in range(world):
        lo, hi = r * V // world, (r + 1) * V // world
        rng = np.random.default_rng(11 * 7919 + r)
        srcs.append(rng.integers(lo, hi, 5000, dtype=np.uint64))
        dsts.append(rng.integers(0, V, 5000, dtype=np.uint64))
    want = PageRank.dense_reference(
        V, np.concatenate(srcs).astype(np.int64),
        np.concatenate(dsts).astype(np.int64), 3)
    np.testing.assert_allclose(got, want, rtol=1e-12)


@pytest.mark.parametrize("world_size", [2, 4, 8])
def test_multiprocess_shuffle(tmp_path, world_size):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_worker,
                         args=(r, world_size, port, str(tmp_path), q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = {}
    for _ in range(world_size):
        rank, payload, remote, local = q.get(timeout=120)
        if isinstance(payload, str):
            raise AssertionError(f"rank {rank} failed: {payload}")
        results[rank] = (np.frombuffer(payload, dtype=np.uint64), remote, local)
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    # global sortedness across partition ranges + completeness
    all_keys = np.concatenate([results[r][0] for r in range(world_size)])
    assert len(all_keys) == world_size * 20000
    boundaries_ok = all(
        results[r][0].max() <= results[r + 1][0].min()
        for r in range(world_size - 1)
        if len(results[r][0]) and len(results[r + 1][0]))
    assert boundaries_ok
    want = np.sort(np.concatenate([
        np.random.default_rng(r).integers(0, 2 ** 32, 20000, dtype=np.uint64)
        for r in range(world_size)]))
    assert np.array_equal(np.sort(all_keys), want)
    # with >1 executors some bytes must have crossed processes
    assert sum(results[r][1] for r in results) > 0


def _crash_worker(rank, world, driver_port, shm_dir, q):
    """Rank `world-1` publishes, wipes its served segments, and dies
    (node-loss simulation); survivors must surface FetchFailedError and
    see the driver prune membership."""
    try:
        import glob
        import sys
        import time
        sys.path.insert(0, os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
        from sparkrdma_amd.conf import ShuffleConf
        from sparkrdma_amd.engine import Engine
        from sparkrdma_amd.partitioner import RangePartitioner
        from sparkrdma_amd.reader import FetchFailedError

        conf = ShuffleConf(shm_dir=shm_dir, max_buffer_allocation_size=1 << 30)
        eng = Engine(conf, rank=rank, world_size=world,
                     driver_port=driver_port)
        R = world * 2
        part = RangePartitioner.uniform(R, key_max=2 ** 32 - 1)
        handle = eng.register_shuffle(num_maps=world, num_partitions=R)
        rng = np.random.default_rng(rank)
        keys = rng.integers(0, 2 ** 32, 5000, dtype=np.uint64)
        w = eng.manager.get_writer(handle, rank)
        w.write_batch(keys, keys.view(np.uint8).reshape(-1, 8).copy())
        w.stop(True, partitioner=part)
        eng.barrier()
        crash_rank = world - 1
        if rank == crash_rank:
            # simulate abrupt node loss AFTER publish: data segments
            # vanish, process dies without any teardown
            for p in glob.glob(os.path.join(
                    shm_dir, f"sparkrdma_{eng.manager.app_id}_e{rank}_s*")):
                if not p.endswith("_s1"):   # keep metadata, kill data
                    os.unlink(p)
            q.put((rank, "crashed"))
            q.close()
            q.join_thread()   # flush before dying (else the msg is lost)
            os._exit(1)
        time.sleep(0.5)   # let the crash + segment wipe land
        try:
            reader = eng.manager.get_reader(handle, rank * 2, rank * 2 + 1)
            list(reader)
            q.put((rank, "NO-ERROR"))
        except FetchFailedError:
            q.put((rank, "fetch-failed"))
        # driver prunes the lost executor and re-announces
        deadline = time.monotonic() + 10
        while time.monotonic() < deadline:
            if crash_rank not in eng.manager._members:
                break
            time.sleep(0.05)
        q.put((rank, "members", sorted(eng.manager._members)))
        eng.manager.stop()
        if eng.driver is not None:
            eng.driver.stop()
    except BaseException as e:
        import traceback
        q.put((rank, f"ERROR: {e}\n{traceback.format_exc()}"))
        raise


def test_executor_crash_mid_shuffle(tmp_path):
    """Node-loss semantics parity: a peer dying after publish fails the
    fetching task (FetchFailedError ≡ Spark FetchFailedException) and the
    driver prunes + re-announces membership (reference
    RdmaShuffleManager.scala:155-165, driver.py finally-block)."""
    import socket
    world = 3
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_crash_worker,
                         args=(r, world, port, str(tmp_path), q))
             for r in range(world)]
    for p in procs:
        p.start()
    msgs = []
    for _ in range(1 + 2 * (world - 1)):   # crash note + 2 per survivor
        msgs.append(q.get(timeout=120))
    for m in msgs:
        assert not (len(m) == 2 and str(m[1]).startswith("ERROR")), m
    outcomes = {m[0]: m[1] for m in msgs if len(m) == 2}
    assert outcomes[world - 1] == "crashed"
    for r in range(world - 1):
        assert outcomes[r] == "fetch-failed", outcomes
    members = {m[0]: m[2] for m in msgs if len(m) == 3}
    for r in range(world - 1):
        assert world - 1 not in members[r], \
            f"rank {r} still sees the dead executor: {members[r]}"
    for p in procs[:world - 1]:
        p.join(timeout=60)
    procs[world - 1].join(timeout=60)


def _tcp_worker(rank, world, driver_port, shm_dir, q):
    try:
        import sys
        sys.path.insert(0, os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
        from sparkrdma_amd.conf import ShuffleConf
        from sparkrdma_amd.engine import Engine
        from sparkrdma_amd.partitioner import HashPartitioner
        from sparkrdma_amd.writer import unpack_partition_segment

        conf = ShuffleConf(shm_dir=shm_dir, transport="tcp",
                           tcp_chunk_size=64 << 10,
                           max_buffer_allocation_size=1 << 30)
        eng = Engine(conf, rank=rank, world_size=world,
                     driver_port=driver_port)
        R = world * 2
        part = HashPartitioner(R)
        handle = eng.register_shuffle(num_maps=world, num_partitions=R)
        rng = np.random.default_rng(100 + rank)
        keys = rng.integers(0, 2 ** 63, 30000, dtype=np.uint64)
        w = eng.manager.get_writer(handle, rank)
        w.write_batch(keys, keys.view(np.uint8).reshape(-1, 8).copy())
        w.stop(True, partitioner=part)
        eng.barrier()
        reader = eng.manager.get_reader(handle, rank * 2, rank * 2 + 1)
        got, remote = [], 0
        for ref, data in reader:
            k, v = unpack_partition_segment(data, 8)
            assert np.array_equal(np.asarray(k).view(np.uint8).reshape(-1),
                                  np.asarray(v).reshape(-1))
            got.append(np.array(k))
        remote = reader.metrics.remote_bytes_read
        csum = int(np.concatenate(got).sum(dtype=np.uint64)) if got else 0
        n = sum(len(g) for g in got)
        q.put((rank, n, csum, remote))
        eng.barrier()
        eng.shutdown()
    except BaseException as e:
        import traceback
        q.put((rank, f"ERROR: {e}\n{traceback.format_exc()}", 0, 0))
        raise


def test_multiprocess_tcp_lane(tmp_path):
    """The cross-host lane with REAL process isolation: every non-self
    read crosses the chunked-streaming data servers (transport=tcp,
    small chunks force multi-chunk responses)."""
    import socket
    world = 3
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_tcp_worker,
                         args=(r, world, port, str(tmp_path), q))
             for r in range(world)]
    for p in procs:
        p.start()
    total_n, total_sum, remote_total = 0, 0, 0
    for _ in range(world):
        rank, n, csum, remote = q.get(timeout=180)
        assert not isinstance(n, str), f"rank {rank}: {n}"
        total_n += n
        total_sum = (total_sum + csum) % (1 << 64)
        remote_total += remote
    for p in procs:
        p.join(timeout=60)
        assert p.exitcode == 0
    assert total_n == world * 30000
    want = 0
    for r in range(world):
        rng = np.random.default_rng(100 + r)
        want = (want + int(rng.integers(0, 2 ** 63, 30000,
                                        dtype=np.uint64).sum(
                                            dtype=np.uint64))) % (1 << 64)
    assert total_sum == want
    assert remote_total > 0, "no bytes crossed the TCP lane"
