"""PageRank over the one-sided shuffle — BASELINE config 4 (the
reference's 19 GB PageRank, README.md:25-31).

Spark-GraphX-style partitioning: edges live partitioned by src (each
executor owns a contiguous vertex range and all edges whose src falls in
it); ranks are co-partitioned, so the per-edge contribution
rank[src]/outdeg[src] is computed locally and one shuffle per iteration
routes (dst, contribution) to dst's owner, which aggregates into the new
rank vector.

Records are the framework's fixed-width pairs: key = dst vertex id (u64),
value = contribution (f64 bit-pattern in the u64 payload).
"""

from __future__ import annotations

import time
from dataclasses import dataclass

import numpy as np

from ..engine import Engine
from ..partitioner import RangePartitioner

DAMPING = 0.85


@dataclass
class PageRankResult:
    seconds: float
    iterations: int
    edges: int
    shuffle_bytes: int


class PageRank:
    def __init__(self, engine: Engine, num_vertices: int,
                 edges_per_executor: int, partitions_per_executor: int = 64,
                 device: str = "cpu", iterations: int = 5, seed: int = 0):
        if num_vertices & (num_vertices - 1):
            raise ValueError("num_vertices must be pow2")
        self.engine = engine
        self.V = num_vertices
        self.iters = iterations
        self.device = device
        W = engine.world_size
        R = W * partitions_per_executor
        if R & (R - 1):
            raise ValueError("total partitions must be pow2")
        self.R = R
        self.ppe = partitions_per_executor
        # partitioner over the vertex id space (span = V, pow2)
        self.part = RangePartitioner.uniform(R, key_min=0, key_max=self.V - 1)
        rank = engine.rank
        self.own_lo = rank * self.V // W
        self.own_hi = (rank + 1) * self.V // W
        span = self.own_hi - self.own_lo
        rng = np.random.default_rng(seed * 7919 + rank)
        src = rng.integers(self.own_lo, self.own_hi, edges_per_executor,
                           dtype=np.uint64)
        dst = rng.integers(0, self.V, edges_per_executor, dtype=np.uint64)
        self.n_edges = edges_per_executor
        outdeg = np.bincount((src - self.own_lo).astype(np.int64),
                             minlength=span).astype(np.float64)
        outdeg[outdeg == 0] = 1.0
        if device == "cuda":
            import torch
            self.src_local = torch.from_numpy(
                (src - self.own_lo).astype(np.int64)).cuda()
            self.dst = torch.from_numpy(dst.view(np.int64)).cuda()
            self.outdeg = torch.from_numpy(outdeg).cuda()
            self.ranks = torch.full((span,), 1.0 / self.V,
                                    dtype=torch.float64, device="cuda")
        else:
            self.src_local = (src - self.own_lo).astype(np.int64)
            self.dst = dst
            self.outdeg = outdeg
            self.ranks = np.full(span, 1.0 / self.V, dtype=np.float64)

    # ------------------------------------------------------------------

    def run_step(self) -> PageRankResult:
        t0 = time.perf_counter()
        shuffle_bytes = 0
        for _ in range(self.iters):
            shuffle_bytes += self._iteration()
        dt = time.perf_counter() - t0
        return PageRankResult(dt, self.iters, self.n_edges, shuffle_bytes)

    def _iteration(self) -> int:
        import os
        dbg = os.environ.get("BENCH_DEBUG")
        t0 = time.perf_counter()
        eng = self.engine
        handle = eng.register_shuffle(eng.world_size, self.R)
        w = eng.manager.get_writer(handle, eng.rank)
        if self.device == "cuda":
            import torch
            contrib = (self.ranks / self.outdeg)[self.src_local]
            w.write_device_batch(self.dst, contrib.view(torch.int64))
        else:
            contrib = (self.ranks / self.outdeg)[self.src_local]
            w.write_batch(self.dst,
                          contrib.view(np.uint8).reshape(-1, 8).copy())
        w.stop(True, partitioner=self.part)
        eng.barrier()
        lo, hi = eng.rank * self.ppe, (eng.rank + 1) * self.ppe - 1
        arena = None
        if self.device == "cuda":
            import torch
            cap = int(self.n_edges * 16 * 1.25) + (64 << 10)
            arena = getattr(self, "_arena_cache", None)
            if arena is None or arena.numel() < cap:
                self._arena_cache = torch.empty(cap, dtype=torch.uint8,
                                                device="cuda")
                arena = self._arena_cache
        reader = eng.manager.get_reader(handle, lo, hi, arena=arena)
        span = self.own_hi - self.own_lo
        # the reader's generic dense keyed-sum (chunks index-add as they
        # arrive, overlapping in-flight fetches) — shared with any
        # iterative workload instead of a per-workload loop
        sums = reader.dense_sum(self.own_lo, span, dtype="f64")
        self.ranks = (1.0 - DAMPING) / self.V + DAMPING * sums
        if self.device == "cuda":
            import torch
            torch.cuda.synchronize()
            if dbg:
                import sys
                print(f"[pr-iter] write+fetch+agg={time.perf_counter()-t0:.3f}s",
                      file=sys.stderr)
        eng.unregister_shuffle(handle)
        return reader.metrics.remote_bytes_read + reader.metrics.local_bytes_read

    # reference implementation for validation (single process, full graph)
    @staticmethod
    def dense_reference(V, src, dst, iterations):
        outdeg = np.bincount(src, minlength=V).astype(np.float64)
        outdeg[outdeg == 0] = 1.0
        ranks = np.full(V, 1.0 / V)
        for _ in range(iterations):
            contrib = ranks / outdeg
            sums = np.zeros(V)
            np.add.at(sums, dst, contrib[src])
            ranks = (1.0 - DAMPING) / V + DAMPING * sums
        return ranks
