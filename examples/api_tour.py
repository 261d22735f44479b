#!/usr/bin/env python3
"""API tour — the ShuffleManager surface in one runnable file (CPU-only).

For a user coming from the reference plugin (Mellanox/SparkRDMA): the same
lifecycle Spark drives through `spark.shuffle.manager` —
registerShuffle → getWriter/write/stop(commit) → getReader/read →
unregisterShuffle — driven directly here. Run:

    python examples/api_tour.py

Everything below works without a GPU (host-shm segments); on an MI355X
box the same code serves map output from HBM and fetches it over xGMI —
the transport is picked per-executor by `transport=auto`.
"""

import os
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np

from sparkrdma_amd.conf import ShuffleConf
from sparkrdma_amd.engine import Engine
from sparkrdma_amd.partitioner import HashPartitioner, RangePartitioner
from sparkrdma_amd.writer import unpack_partition_segment


def main():
    tmp = tempfile.mkdtemp(prefix="sparkrdma_tour_")
    # --- configuration: the reference's spark.shuffle.rdma.* namespace ---
    conf = ShuffleConf.from_dict({
        "spark.shuffle.rdma.maxBytesInFlight": "48m",
        "spark.shuffle.rdma.shuffleReadBlockSize": "256k",
        "spark.shuffle.rdma.shmDir": tmp,
    })

    # --- an Engine is one executor process; rank 0 hosts the registry ---
    # (multi-process: launch N of these under torchrun, one per GPU)
    with Engine(conf, rank=0, world_size=1, driver_port=0) as eng:
        mgr = eng.manager

        # 1) fixed-width records: u64 keys, hash-partitioned -------------
        handle = eng.register_shuffle(num_maps=1, num_partitions=4)
        writer = mgr.get_writer(handle, map_id=0)
        keys = np.arange(100_000, dtype=np.uint64)
        writer.write_batch(keys)                       # stage rows
        writer.stop(True, partitioner=HashPartitioner(4))   # commit+publish
        reader = mgr.get_reader(handle, start_partition=0, end_partition=3)
        got = 0
        for pid, chunks in reader.collect_partitions().items():
            for chunk in chunks:                       # packed segments
                k, _v = unpack_partition_segment(chunk, 0)
                got += len(k)
        assert got == len(keys)
        eng.unregister_shuffle(handle)
        print(f"fixed-width: shuffled {got} u64 keys across 4 partitions")

        # 2) arbitrary Python records: the pickled lane ------------------
        handle = eng.register_shuffle(num_maps=1, num_partitions=2)
        writer = mgr.get_writer(handle, map_id=0)
        writer.write_records([(f"word{i % 10}", 1) for i in range(1000)],
                             partitioner=None)          # default: hash(key)
        writer.stop(True)
        reader = mgr.get_reader(handle, 0, 1)
        counts = {}
        for word, one in reader.read_records():
            counts[word] = counts.get(word, 0) + one
        assert sum(counts.values()) == 1000
        eng.unregister_shuffle(handle)
        print(f"pickled lane: word-counted 1000 records -> {len(counts)} keys")

        # 3) range partitioning (TeraSort-style total order) -------------
        handle = eng.register_shuffle(num_maps=1, num_partitions=8)
        writer = mgr.get_writer(handle, map_id=0)
        rng = np.random.default_rng(0)
        # uniform bounds split the FULL u64 key space (TeraSort convention)
        writer.write_batch(rng.integers(0, 2**64, 50_000, dtype=np.uint64))
        writer.stop(True, partitioner=RangePartitioner.uniform(8))
        reader = mgr.get_reader(handle, 0, 7)
        parts = reader.collect_partitions()            # {pid: [chunks]}
        sizes = [sum(len(unpack_partition_segment(c, 0)[0]) for c in parts[p])
                 for p in sorted(parts)]
        assert all(s > 0 for s in sizes)
        eng.unregister_shuffle(handle)
        print(f"range-partitioned: keys per partition = {sizes}")

        # 4) metrics: the task/executor counters the reference feeds Spark
        print("lifetime:", mgr.lifetime_metrics.format())


if __name__ == "__main__":
    main()
