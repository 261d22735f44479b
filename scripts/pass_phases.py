#!/usr/bin/env python3
"""Per-phase breakdown of the AoS onesweep sort pass (in-kernel
s_memrealtime accumulators): phaseA(load+rank+scan+publish), exchange,
lookback walk (+barrier), writeout. Prints per-block-average us per phase
and the pass wall time — the evidence base for the r02 sort-floor work
(VERDICT item 4)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from sparkrdma_amd.ops import load
from sparkrdma_amd.ops.radix import sort_pairs_aos

REALTIME_MHZ = 100.0   # s_memrealtime ticks at 100 MHz on CDNA


def main():
    m = load()
    n = int(float(sys.argv[1]) * 1e6) if len(sys.argv) > 1 else 64_000_000
    i = torch.arange(n, dtype=torch.int64, device="cuda")
    keys = i * 0x9E3779B97F4A7C15
    keys ^= keys >> 31
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    tmp = torch.empty_like(pairs)
    ws = torch.empty(m.onesweep_workspace_bytes(n, 8), dtype=torch.uint8,
                     device="cuda")
    timing = torch.zeros(4, dtype=torch.int64, device="cuda")

    # warm
    pp = pairs.clone()
    sort_pairs_aos(pp, 0, 64, tmp=tmp, ws=ws)
    torch.cuda.synchronize()

    nblocks = (n + 4095) // 4096
    passes = 8
    names = ["phaseA(load+rank+scan+pub)", "exchange",
             "lookback walk (+barrier)", "writeout"]
    ref = None
    # (lb_mode, split_rounds, lean, label)
    configs = [(0, 0, 0, "baseline (2 blocks/CU)"),
               (0, 2, 1, "lean + split2 (3 blocks/CU)"),
               (0, 4, 1, "lean + split4 (4+ blocks/CU)"),
               (0, 0, 1, "lean only (2 blocks/CU)")]
    import os as _os
    if _os.environ.get("PHASES_TRANSPOSED"):
        configs.append((1, 0, 0, "transposed [ND][nb]"))
    for lb_mode, split, lean, label in configs:
        m.set_lookback_mode(lb_mode)
        m.set_split_exchange(split)
        m.set_lean_pass(lean)
        pp = pairs.clone()
        sort_pairs_aos(pp, 0, 64, tmp=tmp, ws=ws)  # warm this mode
        torch.cuda.synchronize()
        timing.zero_()
        m.set_timing_buf(timing.data_ptr())
        pp = pairs.clone()
        t0 = time.perf_counter()
        out = sort_pairs_aos(pp, 0, 64, tmp=tmp, ws=ws)
        torch.cuda.synchronize()
        wall = time.perf_counter() - t0
        m.set_timing_buf(0)
        keys_sorted = out[0::2]
        if ref is None:
            ref = keys_sorted.clone()
        else:
            assert torch.equal(ref, keys_sorted), "modes disagree!"
        # timed run again without the timing instrumentation
        pp = pairs.clone()
        t0 = time.perf_counter()
        sort_pairs_aos(pp, 0, 64, tmp=tmp, ws=ws)
        torch.cuda.synchronize()
        wall_clean = time.perf_counter() - t0
        t = timing.cpu().numpy() / REALTIME_MHZ / (nblocks * passes)
        print(f"{label}: wall={wall_clean*1e3:.2f} ms "
              f"({wall_clean/passes*1e3:.3f} ms/pass; instrumented "
              f"{wall/passes*1e3:.3f})")
        for nm, v in zip(names, t):
            print(f"  {nm:<28} {v:7.2f} us/block")
        print(f"  {'total in-kernel':<28} {t.sum():7.2f} us/block")
    m.set_lookback_mode(0)
    m.set_split_exchange(0)
    m.set_lean_pass(0)


if __name__ == "__main__":
    main()
