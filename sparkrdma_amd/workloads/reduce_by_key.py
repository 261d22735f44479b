"""reduceByKey — the aggregation role of the shuffle reader
(reference RdmaShuffleReader.scala:61-114: deserialize → aggregate),
realized on GPU.

Synthetic (key u64, value u64) pairs with a bounded key space; one step:
hash-partition by splitmix64 (the kernel's hash_mix digit — bit-identical
to HashPartitioner), one-sided shuffle, then per-rank: AoS sort by key
and segmented sum over equal-key runs. Validated against a CPU
dictionary oracle.
"""

from __future__ import annotations

import time
from dataclasses import dataclass

import numpy as np

from ..engine import Engine
from ..partitioner import HashPartitioner


@dataclass
class ReduceByKeyResult:
    seconds: float
    rows: int
    groups: int
    shuffle_bytes: int


class ReduceByKey:
    def __init__(self, engine: Engine, rows_per_executor: int,
                 num_keys: int = 1 << 20, partitions_per_executor: int = 32,
                 device: str = "cpu", validate: bool = False, seed: int = 0):
        self.engine = engine
        self.n = rows_per_executor
        self.device = device
        self.validate = validate
        W = engine.world_size
        R = W * partitions_per_executor
        if R & (R - 1):
            raise ValueError("total partitions must be pow2")
        self.R = R
        self.ppe = partitions_per_executor
        self.part = HashPartitioner(R)
        self.seed = seed
        self.num_keys = num_keys
        rng = np.random.default_rng(seed * 101 + engine.rank)
        k = rng.integers(0, num_keys, self.n, dtype=np.uint64)
        v = rng.integers(0, 1 << 20, self.n, dtype=np.uint64)
        self._k_np, self._v_np = k, v
        if device == "cuda":
            import torch
            self.keys = torch.from_numpy(k.view(np.int64)).cuda()
            self.vals = torch.from_numpy(v.view(np.int64)).cuda()
        else:
            self.keys = k
            self.vals = v.view(np.uint8).reshape(-1, 8).copy()

    def run_step(self) -> ReduceByKeyResult:
        eng = self.engine
        t0 = time.perf_counter()
        handle = eng.register_shuffle(eng.world_size, self.R)
        w = eng.manager.get_writer(handle, eng.rank)
        if self.device == "cuda":
            w.write_device_batch(self.keys, self.vals)
        else:
            w.write_batch(self.keys, self.vals)
        w.stop(True, partitioner=self.part)
        eng.barrier()
        lo, hi = eng.rank * self.ppe, (eng.rank + 1) * self.ppe - 1
        reader = eng.manager.get_reader(handle, lo, hi)
        # the reader's GENERIC aggregate hookup (RdmaShuffleReader.scala:
        # 61-114 role): sort by key + segmented sum, device-side on GPU
        uk, sums = reader.read_aos(aggregator="sum")
        if self.device == "cuda":
            import torch
            torch.cuda.synchronize()
            groups = int(uk.numel())
        else:
            groups = len(uk)
        if self.validate:
            self._validate(uk, sums)
        eng.unregister_shuffle(handle)
        return ReduceByKeyResult(
            time.perf_counter() - t0, self.n, groups,
            reader.metrics.remote_bytes_read + reader.metrics.local_bytes_read)

    def _validate(self, uk, sums) -> None:
        """Single-rank oracle: this rank's groups must equal the dict-based
        sums of every rank's rows hashing to this rank's partitions."""
        eng = self.engine
        want: dict = {}
        for r in range(eng.world_size):
            # regenerate rank r's rows deterministically
            rng = np.random.default_rng(self.seed * 101 + r)
            k = rng.integers(0, self.num_keys, self.n, dtype=np.uint64)
            v = rng.integers(0, 1 << 20, self.n, dtype=np.uint64)
            pids = self.part.partition_ids(k)
            mine = (pids >= eng.rank * self.ppe) & \
                   (pids < (eng.rank + 1) * self.ppe)
            for kk, vv in zip(k[mine], v[mine]):
                want[int(kk)] = want.get(int(kk), 0) + int(vv)
        if self.device == "cuda":
            uk = uk.cpu().numpy().view(np.uint64)
            sums = sums.cpu().numpy().view(np.uint64)
        got = dict(zip(uk.tolist(), sums.tolist()))
        assert got == want, (len(got), len(want))
