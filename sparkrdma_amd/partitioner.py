"""Partitioners: key -> reduce-partition id.

The reference delegates partitioning to Spark's Partitioner inside the
sort-shuffle writers (RdmaWrapperShuffleWriter.scala:83-102). The rebuild
owns it: partition-id computation is vectorized (numpy on host, HIP kernel
on GPU — ops/hipshuffle kernels use the same functions bit-for-bit).
"""

from __future__ import annotations

import numpy as np


class HashPartitioner:
    """Multiplicative-mix hash of uint64 keys, modulo num_partitions.

    The mix constant is splitmix64's (0xff51afd7ed558ccd finalizer step) —
    the GPU kernel in ops/csrc/kernels.hip implements the identical
    function so CPU oracle tests compare bit-for-bit.
    """

    MIX = np.uint64(0xFF51AFD7ED558CCD)

    def __init__(self, num_partitions: int):
        self.num_partitions = num_partitions
        # GPU fast path: for pow2 R >= 16, digit = hash_mix64(k) & (R-1)
        # in the kernel is bit-identical to partition_ids (% == & for pow2)
        nbits = (num_partitions - 1).bit_length()
        if (1 << nbits) == num_partitions and 4 <= nbits <= 12:
            self.gpu_hash = True
            self.gpu_shift = 0

    def partition_ids(self, keys: np.ndarray) -> np.ndarray:
        k = keys.astype(np.uint64, copy=False)
        with np.errstate(over="ignore"):
            h = (k ^ (k >> np.uint64(33))) * self.MIX
            h ^= h >> np.uint64(33)
        return (h % np.uint64(self.num_partitions)).astype(np.int32)


class RangePartitioner:
    """Range partitioner over uint64 keys — TeraSort's partitioner.

    ``bounds`` are num_partitions-1 ascending split points; partition i
    holds keys in [bounds[i-1], bounds[i]).
    """

    def __init__(self, bounds: np.ndarray):
        self.bounds = np.asarray(bounds, dtype=np.uint64)
        self.num_partitions = len(self.bounds) + 1

    @classmethod
    def uniform(cls, num_partitions: int,
                key_min: int = 0, key_max: int = 2 ** 64 - 1) -> "RangePartitioner":
        span = (key_max - key_min + 1) if key_max < 2 ** 64 - 1 else 2 ** 64
        bounds = [key_min + (span * (i + 1)) // num_partitions
                  for i in range(num_partitions - 1)]
        p = cls(np.array(bounds, dtype=np.uint64))
        # GPU fast path: pow2 partitions over a pow2 key span starting at 0
        # => partition id is a plain top-bit shift the radix kernel can use.
        nbits = (num_partitions - 1).bit_length()
        if (key_min == 0 and (1 << nbits) == num_partitions
                and span & (span - 1) == 0):
            p.gpu_shift = span.bit_length() - 1 - nbits
        return p

    def partition_ids(self, keys: np.ndarray) -> np.ndarray:
        k = keys.astype(np.uint64, copy=False)
        return np.searchsorted(self.bounds, k, side="right").astype(np.int32)
