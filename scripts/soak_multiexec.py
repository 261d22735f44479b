#!/usr/bin/env python3
"""Two-executor control-plane soak: every cycle, both executors write a
map output and each reads the OTHER's partitions too (cross-executor
one-sided reads), verifying counts — endurance for the announce/prebuild/
remote-read lane that the single-executor soak cannot exercise.

  python scripts/soak_multiexec.py [minutes] [seed] [transport] [world]

transport "tcp" forces every cross-executor read through the data
servers (the multi-node lane) — endurance for the streaming protocol
and the client connection pool.
"""

import multiprocessing as mp
import os
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def worker(rank, world, driver_port, shm_dir, minutes, seed, q,
           transport="auto"):
    import random
    import numpy as np
    from sparkrdma_amd.conf import ShuffleConf
    from sparkrdma_amd.engine import Engine
    from sparkrdma_amd.partitioner import HashPartitioner

    rng = random.Random(seed * 100 + 7)   # SAME stream on both ranks
    conf = ShuffleConf(shm_dir=shm_dir, max_buffer_allocation_size=1 << 30,
                       transport=transport)
    rounds = recs = 0
    try:
        with Engine(conf, rank=rank, world_size=world,
                    driver_port=driver_port) as eng:
            deadline = time.monotonic() + minutes * 60
            # termination: rank 0 owns the clock; other ranks follow until
            # rank 0 departs (its shutdown releases the barrier, then the
            # next lookup fails LOUDLY with KeyError/ConnectionError — the
            # framework's desync detection doubling as the stop signal)
            while rank != 0 or time.monotonic() < deadline:
                R = world * rng.choice((1, 2, 4, 8))  # same on all ranks
                n = rng.randrange(100, 20_000)
                part = HashPartitioner(R)
                try:
                    h = eng.register_shuffle(world, R)
                    w = eng.manager.get_writer(h, rank)
                    keys = np.random.default_rng(
                        seed + rounds * world + rank).integers(
                        0, 2 ** 64, n, dtype=np.uint64)
                    w.write_batch(keys)
                    w.stop(True, partitioner=part)
                    eng.barrier()
                    lo = rank * (R // world)
                    hi = (rank + 1) * (R // world) - 1
                    reader = eng.manager.get_reader(h, lo, hi)
                    got = sum(len(c) // 8 for _ref, c in reader)
                    # expected: both ranks' keys falling into [lo, hi]
                    want = 0
                    for r2 in range(world):
                        k2 = np.random.default_rng(
                            seed + rounds * world + r2).integers(
                            0, 2 ** 64, n, dtype=np.uint64)
                        pid = part.partition_ids(k2)
                        want += int(np.sum((pid >= lo) & (pid <= hi)))
                    assert got == want, \
                        f"rank {rank} round {rounds}: {got} != {want}"
                    assert reader.metrics.remote_blocks_fetched > 0 or world == 1
                    eng.unregister_shuffle(h)
                except AssertionError:
                    raise       # correctness failures are NEVER clean stops
                except Exception:
                    if rank != 0 and time.monotonic() > deadline - 30:
                        break   # rank 0 exited mid-round: clean stop
                    raise
                rounds += 1
                recs += n
        q.put((rank, rounds, recs, None))
    except BaseException as e:
        q.put((rank, rounds, recs, repr(e)))
        raise


def main():
    minutes = float(sys.argv[1]) if len(sys.argv) > 1 else 10
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    transport = sys.argv[3] if len(sys.argv) > 3 else "auto"
    tmp = tempfile.mkdtemp(prefix="sparkrdma_soak2_")
    import socket
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    world = int(sys.argv[4]) if len(sys.argv) > 4 else 2
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=worker,
                         args=(r, world, port, tmp, minutes, seed, q,
                               transport))
             for r in range(world)]
    for p in procs:
        p.start()
    fails = []
    for _ in range(world):
        rank, rounds, recs, err = q.get(timeout=minutes * 60 + 120)
        print(f"rank {rank}: {rounds} shuffles, {recs} records, err={err}",
              flush=True)
        if err:
            fails.append((rank, err))
    for p in procs:
        p.join(timeout=60)
    if fails:
        print("SOAK FAILED:", fails)
        sys.exit(1)
    print(f"multi-executor soak ok ({minutes:.0f} min, world={world}, "
          f"transport={transport}, cross-executor reads verified every cycle)")


if __name__ == "__main__":
    main()
