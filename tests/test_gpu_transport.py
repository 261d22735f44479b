"""Round-2 transport mechanics on a real MI355X: non-blocking issue depth
(event-driven completions), event recycling, and
announce-time slab pre-import (item 1)."""

import multiprocessing as mp
import os
import socket
import sys
import time

import pytest

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hs():
    from sparkrdma_amd.ops import load
    m = load()
    m.set_device(0)
    return m


def test_issue_depth_not_thread_bound(hs):
    """Enqueue 64 one-sided copy batches WITHOUT waiting on any: issue is
    non-blocking, so in-flight depth is set by flow control, not by a
    count of blocked threads (the r01 model capped depth at 8 GIL-bound
    workers)."""
    import torch
    n = 1 << 20
    src = torch.arange(64 * n, dtype=torch.int64, device="cuda")
    dst = torch.empty_like(src)
    evs = []
    t0 = time.perf_counter()
    for i in range(64):
        evs.append(hs.read_batch(0, [dst.data_ptr() + i * n * 8],
                                 [src.data_ptr() + i * n * 8], [n * 8]))
    issue_s = time.perf_counter() - t0
    # pure enqueue: far faster than the copies themselves
    assert issue_s < 0.5, issue_s
    deadline = time.monotonic() + 30
    pending = set(evs)
    while pending and time.monotonic() < deadline:
        pending = {e for e in pending if not hs.poll_event(e)}
        time.sleep(0.0005)
    assert not pending
    assert torch.equal(src, dst)


def test_event_recycling_many_batches(hs):
    """Thousands of batches reuse the recycled event free-list (no
    create/destroy churn, no leak-driven failure)."""
    import torch
    src = torch.ones(1024, dtype=torch.uint8, device="cuda")
    dst = torch.empty_like(src)
    for round_ in range(50):
        evs = [hs.read_batch(round_ % 8, [dst.data_ptr()],
                             [src.data_ptr()], [1024])
               for _ in range(40)]
        for e in evs:
            hs.wait_event(e)
    assert torch.equal(src, dst)


def _prebuild_worker(rank, world, driver_port, shm_dir, q):
    try:
        sys.path.insert(0, os.path.dirname(os.path.dirname(
            os.path.abspath(__file__))))
        import torch
        from sparkrdma_amd.conf import ShuffleConf
        from sparkrdma_amd.engine import Engine
        from sparkrdma_amd.partitioner import RangePartitioner

        conf = ShuffleConf(transport="ipc", hbm_pool_size=1 << 30,
                           shm_dir=shm_dir, gpu_id=0)
        eng = Engine(conf, rank=rank, world_size=world,
                     driver_port=driver_port)
        R = 8
        part = RangePartitioner.uniform(R)
        handle = eng.register_shuffle(world, R)
        keys = torch.randint(-2**63, 2**63 - 1, (100_000,),
                             dtype=torch.int64, device="cuda")
        w = eng.manager.get_writer(handle, rank)
        w.write_device_batch(keys, keys.clone())
        w.stop(True, partitioner=part)
        eng.barrier()   # all map outputs (and their slabs) published
        # the announce-triggered prebuild imports peers' published slab
        # handles in the background; wait for it rather than fetching
        deadline = time.monotonic() + 10
        peer_keys = []
        while time.monotonic() < deadline:
            peer_keys = [k for k in eng.manager.gpu._peer_bases
                         if (k >> 16) != rank]
            if peer_keys:
                break
            time.sleep(0.05)
        q.put((rank, len(peer_keys)))
        # fetch must still work (and hit the pre-imported base)
        lo = rank * (R // world)
        reader = eng.manager.get_reader(handle, lo, lo + R // world - 1)
        n_blocks = sum(1 for _ in reader)
        q.put((rank, "blocks", n_blocks))
        eng.barrier()
        eng.shutdown()
    except BaseException as e:
        import traceback
        q.put((rank, f"ERROR: {e}\n{traceback.format_exc()}"))
        raise


def test_prebuild_imports_peer_slabs(tmp_path):
    """After a peer publishes its map output, the background pre-build
    imports its slab IPC handles BEFORE any fetch touches them — hop-3
    issue latency is then independent of first-touch (VERDICT r01 done
    criterion for item 1)."""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    procs = [ctx.Process(target=_prebuild_worker,
                         args=(r, 2, port, str(tmp_path), q))
             for r in range(2)]
    for p in procs:
        p.start()
    msgs = []
    for _ in range(4):
        msgs.append(q.get(timeout=300))
    for m in msgs:
        assert not (len(m) == 2 and isinstance(m[1], str)), m
    pre_imports = {m[0]: m[1] for m in msgs if len(m) == 2}
    blocks = {m[0]: m[2] for m in msgs if len(m) == 3}
    for p in procs:
        p.join(timeout=120)
        assert p.exitcode == 0
    for rank in (0, 1):
        assert pre_imports[rank] >= 1, \
            f"rank {rank}: no peer slabs imported by prebuild"
        assert blocks[rank] > 0
