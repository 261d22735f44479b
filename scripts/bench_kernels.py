"""Micro-bench of the radix kernels on one MI355X.

Run on a GPU box:  python scripts/bench_kernels.py [n_million]
Prints partition + sort throughput (records/s and effective GB/s).
"""

import os
import sys
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sparkrdma_amd.ops import load
from sparkrdma_amd.ops.radix import radix_partition, sort_pairs


def bench(fn, iters=5, warmup=2):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    n = int(float(sys.argv[1]) * 1e6) if len(sys.argv) > 1 else 64_000_000
    load()
    torch.manual_seed(0)
    keys = torch.randint(-2**63, 2**63 - 1, (n,), dtype=torch.int64,
                         device="cuda")
    vals = torch.randint(-2**63, 2**63 - 1, (n,), dtype=torch.int64,
                         device="cuda")
    rec_bytes = 16

    for nbits in (8, 10, 11):
        t = bench(lambda: radix_partition(keys, vals, nbits))
        print(f"partition nbits={nbits}: {t*1e3:8.2f} ms  "
              f"{n/t/1e9:6.2f} Grec/s  {n*rec_bytes/t/1e9:7.1f} GB/s payload")

    for passes, end_bit in ((4, 32), (7, 56), (8, 64)):
        for osweep in (False, True):
            kk = keys.clone()
            vv = vals.clone()
            t = bench(lambda: sort_pairs(kk, vv, 0, end_bit, onesweep=osweep))
            tag = "onesweep" if osweep else "3-kernel"
            print(f"sort {passes}p {tag} (bits 0..{end_bit}): {t*1e3:8.2f} ms  "
                  f"{n/t/1e9:6.2f} Grec/s  {n*rec_bytes/t/1e9:7.1f} GB/s payload")

    from sparkrdma_amd.ops.radix import sort_pairs_aos
    pairs = torch.stack([keys, vals], dim=1).reshape(-1).contiguous()
    for passes, end_bit in ((7, 56), (8, 64)):
        pp = pairs.clone()
        t = bench(lambda: sort_pairs_aos(pp, 0, end_bit))
        print(f"sort {passes}p AoS-onesweep (bits 0..{end_bit}): {t*1e3:8.2f} ms  "
              f"{n/t/1e9:6.2f} Grec/s  {n*rec_bytes/t/1e9:7.1f} GB/s payload")

    # torch baseline for context
    t = bench(lambda: torch.sort(keys)[0])
    print(f"torch.sort keys-only baseline: {t*1e3:8.2f} ms  {n/t/1e9:6.2f} Grec/s")


if __name__ == "__main__":
    main()

def tile_ab():
    import torch
    from sparkrdma_amd.ops import load
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    m = load()
    n = 64_000_000
    keys = torch.randint(-2**63, 2**63 - 1, (n,), dtype=torch.int64, device="cuda")
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    for tile in (2048, 4096, 8192):
        m.set_aos_tile(tile)
        pp = pairs.clone()
        t = bench(lambda: sort_pairs_aos(pp, 0, 64))
        print(f"AoS tile {tile}: {t*1e3:8.2f} ms  {n*16/t/1e9:7.1f} GB/s")
    m.set_aos_tile(4096)

if __name__ == "__main__" and os.environ.get("TILE_AB"):
    tile_ab()

def digit_ab():
    import torch
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    n = 64_000_000
    keys = torch.randint(-2**63, 2**63 - 1, (n,), dtype=torch.int64, device="cuda")
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    for bits_range in (56, 64):
        for db in (7, 8):
            pp = pairs.clone()
            t = bench(lambda: sort_pairs_aos(pp, 0, bits_range, digit_bits=db))
            print(f"AoS {bits_range}b digit={db}: {t*1e3:8.2f} ms  {n*16/t/1e9:7.1f} GB/s")

if __name__ == "__main__" and os.environ.get("DIGIT_AB"):
    digit_ab()

def stage_ab():
    import torch, sys
    from sparkrdma_amd.ops import load
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    m = load()
    n = int(float(sys.argv[1]) * 1e6) if len(sys.argv) > 1 else 64_000_000
    i = torch.arange(n, dtype=torch.int64, device="cuda")
    keys = i * 0x9E3779B97F4A7C15
    keys ^= keys >> 31
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    for stage, label in ((9, "no LDS counter chain (timing only)"),
                         (0, "rank only, no lookback walk"),
                         (1, "rank+lookback only"), (2, "+exch+dummy-store"),
                         (3, "full")):
        m.set_pass_stage(stage)
        pp = pairs.clone()
        t = bench(lambda: sort_pairs_aos(pp, 0, 64))
        print(f"stage {stage} ({label}): {t*1e3:8.2f} ms  ({t/8*1e3:.3f} ms/pass)")
    m.set_pass_stage(3)

if __name__ == "__main__" and os.environ.get("STAGE_AB"):
    stage_ab()

def timing_probe():
    import torch
    from sparkrdma_amd.ops import load
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    m = load()
    n = int(float(sys.argv[1]) * 1e6) if len(sys.argv) > 1 else 64_000_000
    i = torch.arange(n, dtype=torch.int64, device="cuda")
    keys = i * 0x9E3779B97F4A7C15
    keys ^= keys >> 31
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    tbuf = torch.zeros(4, dtype=torch.int64, device="cuda")
    m.set_timing_buf(tbuf.data_ptr())
    pp = pairs.clone()
    t = bench(lambda: sort_pairs_aos(pp, 0, 64), iters=1, warmup=1)
    m.set_timing_buf(0)
    torch.cuda.synchronize()
    vals = tbuf.cpu().numpy() / 2  # 2 runs (warmup+timed)
    nb = (n + 4095) // 4096
    per_block_us = vals / (nb * 8) / 100.0  # 100 MHz realtime clock, 8 passes
    names = ["phaseA+scan+publish", "exchange", "lookback+walk", "writeout"]
    print(f"full sort: {t*1e3:.2f} ms; per-block phase means (us):")
    for nm, v in zip(names, per_block_us):
        print(f"  {nm:22s} {v:8.2f} us")

if __name__ == "__main__" and os.environ.get("TIMING_PROBE"):
    timing_probe()

def mode_ab():
    import torch
    from sparkrdma_amd.ops import load
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    m = load()
    n = int(float(sys.argv[1]) * 1e6) if len(sys.argv) > 1 else 64_000_000
    i = torch.arange(n, dtype=torch.int64, device="cuda")
    keys = i * 0x9E3779B97F4A7C15
    keys ^= keys >> 31
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    ref = None
    for mode, label in ((0, "lookback"), (1, "hist+scan")):
        m.set_sort_mode(mode)
        pp = pairs.clone()
        t = bench(lambda: sort_pairs_aos(pp, 0, 64))
        out = sort_pairs_aos(pairs.clone(), 0, 64)
        torch.cuda.synchronize()
        if ref is None:
            ref = out
        else:
            assert torch.equal(out, ref), f"mode {mode} output differs!"
        print(f"mode {mode} ({label}): {t*1e3:8.2f} ms")
    m.set_sort_mode(1)

if __name__ == "__main__" and os.environ.get("MODE_AB"):
    mode_ab()
