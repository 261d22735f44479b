"""Partitioners: key -> reduce-partition id.

The reference delegates partitioning to Spark's Partitioner inside the
sort-shuffle writers (RdmaWrapperShuffleWriter.scala:83-102). The rebuild
owns it: partition-id computation is vectorized (numpy on host, HIP kernel
on GPU — ops/hipshuffle kernels use the same functions bit-for-bit).

GPU dispatch: ``gpu_params()`` returns (func, shift, nparts) selecting the
kernel-side partition function (kernels.hip ``PartFunc``):
  0 bits       (key >> shift) & (R-1)        pow2 R
  1 hash+bits  (mix(key) >> shift) & (R-1)   pow2 R >= 16
  2 range      mulhi(key, R)                 ANY R  (uniform u64 range)
  3 hash mod   mix(key) % R                  ANY R
Funcs 2/3 lift the r01 pow2-only restriction. The LDS counter layout
caps ONE kernel pass at 2^12 digits; the writer runs a two-level
coarse/fine pid radix for 4096 < R <= 2^24 (writer._commit_gpu_records).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np

GPU_MAX_PARTITIONS = 1 << 24


class HashPartitioner:
    """Multiplicative-mix hash of uint64 keys, modulo num_partitions.

    The mix constant is splitmix64's (0xff51afd7ed558ccd finalizer step) —
    the GPU kernel in ops/csrc/kernels.hip implements the identical
    function so CPU oracle tests compare bit-for-bit.
    """

    MIX = np.uint64(0xFF51AFD7ED558CCD)

    def __init__(self, num_partitions: int):
        self.num_partitions = num_partitions

    def gpu_params(self) -> Optional[Tuple[int, int, int]]:
        R = self.num_partitions
        nbits = (R - 1).bit_length()
        if (1 << nbits) == R and 4 <= nbits <= 12:
            return (1, 0, 0)        # hash then & (R-1) == % R for pow2
        return (3, 0, R)            # hash then % R (any R; the writer
                                    # two-levels when R > 4096)

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
        self._uniform_full_range = False

    @classmethod
    def uniform(cls, num_partitions: int,
                key_min: int = 0, key_max: int = 2 ** 64 - 1) -> "RangePartitioner":
        span = (key_max - key_min + 1) if key_max < 2 ** 64 - 1 else 2 ** 64
        # CEILING bounds: partition(key) == floor(key * R / span), exactly
        # the kernel's multiply-high (func 2) for the full-u64 case — so
        # GPU and CPU agree at the boundary keys bit-for-bit
        bounds = [key_min + -(-span * (i + 1) // num_partitions)
                  for i in range(num_partitions - 1)]
        p = cls(np.array(bounds, dtype=np.uint64))
        # pow2 partitions over a pow2 key span starting at 0 => partition
        # id is a plain top-bit shift (func 0)
        nbits = (num_partitions - 1).bit_length()
        if (key_min == 0 and (1 << nbits) == num_partitions
                and span & (span - 1) == 0):
            p.gpu_shift = span.bit_length() - 1 - nbits
        p._uniform_full_range = (key_min == 0 and span == 2 ** 64)
        return p

    def gpu_params(self) -> Optional[Tuple[int, int, int]]:
        R = self.num_partitions
        if hasattr(self, "gpu_shift"):
            return (0, self.gpu_shift, 0)
        if self._uniform_full_range:
            return (2, 0, R)        # mulhi(key, R) — any R; the writer
                                    # two-levels when R > 4096
        return None                 # custom bounds: CPU path only

    def partition_ids(self, keys: np.ndarray) -> np.ndarray:
        k = keys.astype(np.uint64, copy=False)
        return np.searchsorted(self.bounds, k, side="right").astype(np.int32)
