"""Deterministic sort-pass probe for PMC runs (no curand kernels).

Generates pseudo-random u64 keys with arithmetic only, then runs the AoS
onesweep sort a few times. Use under:
  rocprofv3 --pmc FETCH_SIZE WRITE_SIZE -- python scripts/prof_pass.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sparkrdma_amd.ops import load
from sparkrdma_amd.ops.radix import sort_pairs, sort_pairs_aos

load()
n = int(float(sys.argv[1]) * 1e6) if len(sys.argv) > 1 else 64_000_000
i = torch.arange(n, dtype=torch.int64, device="cuda")
keys = i * 0x9E3779B97F4A7C15  # golden-ratio LCG: uniform-ish digit spread
keys ^= keys >> 31
pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()


def t(fn, iters=3):
    fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


kk = keys.clone()
d = t(lambda: sort_pairs(kk, None, 0, 64))          # keys-only SoA: r8+w8
print(f"keys-only SoA 8p: {d*1e3:8.2f} ms  ({d/8*1e3:.3f} ms/pass)")
kk = keys.clone()
vv = keys.clone()
d = t(lambda: sort_pairs(kk, vv, 0, 64))            # SoA pairs: r24+w16
print(f"pairs SoA 8p:     {d*1e3:8.2f} ms  ({d/8*1e3:.3f} ms/pass)")
pp = pairs.clone()
d = t(lambda: sort_pairs_aos(pp, 0, 64))            # AoS pairs: r16+w16
print(f"pairs AoS 8p:     {d*1e3:8.2f} ms  ({d/8*1e3:.3f} ms/pass)")
