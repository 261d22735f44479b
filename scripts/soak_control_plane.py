#!/usr/bin/env python3
"""CPU control-plane soak: randomized shuffles through the full engine
(register → write → read → verify → unregister) for N minutes, mixing
record shapes, partitioners, partition counts and record counts — the
long-running-executor endurance check (id/metadata recycling under load).

  python scripts/soak_control_plane.py [minutes] [seed]
"""

import os
import random
import sys
import tempfile
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.engine import Engine
from sparkrdma_amd.partitioner import HashPartitioner, RangePartitioner
from sparkrdma_amd.writer import unpack_partition_segment


def one_round(eng, rng):
    R = rng.choice((2, 3, 4, 7, 8, 16, 64, 100))
    n = rng.randrange(1, 50_000)
    handle = eng.register_shuffle(1, R)
    w = eng.manager.get_writer(handle, 0)
    kind = rng.random()
    if kind < 0.45:                      # fixed-width u64 keys (+values)
        keys = np.random.default_rng(rng.getrandbits(32)).integers(
            0, 2 ** 64, n, dtype=np.uint64)
        vals = keys.view(np.uint8).reshape(-1, 8).copy() \
            if rng.random() < 0.5 else None
        w.write_batch(keys, vals)
        part = (RangePartitioner.uniform(R) if rng.random() < 0.5
                else HashPartitioner(R))
        w.stop(True, partitioner=part)
        reader = eng.manager.get_reader(handle, 0, R - 1)
        got = 0
        for p, chunks in reader.collect_partitions().items():
            for c in chunks:
                k, v = unpack_partition_segment(c, 8 if vals is not None else 0)
                got += len(k)
                pid = part.partition_ids(np.asarray(k))
                assert np.all(pid == p), "record in wrong partition"
        assert got == n, f"lost records {got}/{n}"
    else:                                # pickled python records
        w.write_records(((rng.getrandbits(40), i) for i in range(n)), None)
        w.stop(True)
        reader = eng.manager.get_reader(handle, 0, R - 1)
        cnt = sum(1 for _ in reader.read_records())
        assert cnt == n, f"lost pickled records {cnt}/{n}"
    eng.unregister_shuffle(handle)
    return n


def main():
    minutes = float(sys.argv[1]) if len(sys.argv) > 1 else 10
    seed = int(sys.argv[2]) if len(sys.argv) > 2 else 0
    rng = random.Random(seed)
    tmp = tempfile.mkdtemp(prefix="sparkrdma_soak_")
    conf = ShuffleConf(shm_dir=tmp, max_buffer_allocation_size=1 << 30)
    deadline = time.monotonic() + minutes * 60
    rounds = recs = 0
    with Engine(conf, rank=0, world_size=1, driver_port=0) as eng:
        while time.monotonic() < deadline:
            recs += one_round(eng, rng)
            rounds += 1
            if rounds % 100 == 0:
                print(f"  {rounds} shuffles, {recs} records, "
                      f"{deadline - time.monotonic():.0f}s left", flush=True)
        print(f"soak ok: {rounds} randomized shuffles, {recs} records, "
              f"{minutes:.0f} min, metadata/ids recycled throughout")


if __name__ == "__main__":
    main()
