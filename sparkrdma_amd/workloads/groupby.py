"""groupByKey over arbitrary pickled records — BASELINE config 1
(Spark local-cluster[2,1,1024] groupByKey on 1M synthetic rows; the
plumbing path, no GPU required).

Exercises the bytes-record write path, hash partitioning by pickled key,
and stream deserialization on the reduce side — the closest analog of the
reference's production validation workload shape.
"""

from __future__ import annotations

import time
from dataclasses import dataclass

import numpy as np

from ..engine import Engine


@dataclass
class GroupByResult:
    seconds: float
    rows: int
    groups: int


class GroupByKey:
    def __init__(self, engine: Engine, rows_per_executor: int,
                 num_keys: int = 1000, num_partitions: int = 0, seed: int = 0):
        self.engine = engine
        self.n = rows_per_executor
        self.num_keys = num_keys
        self.R = num_partitions or engine.world_size * 4
        rng = np.random.default_rng(seed + engine.rank)
        self.keys = rng.integers(0, num_keys, rows_per_executor)

    def run_step(self) -> GroupByResult:
        eng = self.engine
        t0 = time.perf_counter()
        handle = eng.register_shuffle(eng.world_size, self.R)
        w = eng.manager.get_writer(handle, eng.rank)
        w.write_records(((f"key{k}", (eng.rank, int(k))) for k in self.keys),
                        None)
        w.stop(True)
        eng.barrier()
        per = self.R // eng.world_size
        lo, hi = eng.rank * per, (eng.rank + 1) * per - 1
        reader = eng.manager.get_reader(handle, lo, hi)
        groups = {}
        # reader-side generic deserialize (read_records) + group-by
        for k, v in reader.read_records():
            groups.setdefault(k, []).append(v)
        eng.unregister_shuffle(handle)
        return GroupByResult(time.perf_counter() - t0, self.n, len(groups))
