"""Torch-facing wrappers over the CDNA4 radix kernels.

Fail-loud policy: on a CUDA(ROCm)-visible machine these functions REQUIRE
the native extension; there is no silent eager fallback (CPU oracles live
in tests, not here).
"""

from __future__ import annotations

from typing import Optional, Tuple

import torch

from . import load


def _stream() -> int:
    return torch.cuda.current_stream().cuda_stream


def radix_partition(keys: torch.Tensor, vals: Optional[torch.Tensor],
                    nbits: int, shift: Optional[int] = None,
                    key_dst: Optional[torch.Tensor] = None,
                    val_dst: Optional[torch.Tensor] = None,
                    hash_mix: bool = False
                    ) -> Tuple[torch.Tensor, torch.Tensor, Optional[torch.Tensor]]:
    """One bucket-scatter pass: digit = (key >> shift) & (2^nbits - 1).

    If key_dst/val_dst (u64 device-pointer tables, one per bucket) are
    given, elements scatter straight to those addresses (the map-side
    "write into HBM block buffers" path) and only ``counts`` is returned
    meaningfully. Otherwise contiguous outputs are allocated and per-digit
    bases derived from the counts.

    Returns (counts u32[2^nbits] on device, keys_out, vals_out).
    """
    m = load()
    n = keys.numel()
    assert keys.dtype in (torch.int64, torch.uint64), "keys must be 64-bit"
    if shift is None:
        shift = 64 - nbits
    nbits_eff = max(nbits, 4)  # kernel instantiations start at 4 bits; the
    nd = 1 << nbits_eff        # extra mask bits read zeros above the digit
    dev = keys.device
    hist = torch.empty(m.radix_hist_bytes(n, nbits_eff) // 4,
                       dtype=torch.int32, device=dev)
    scan_ws = torch.empty(m.radix_scan_ws_bytes(n, nbits_eff) // 4,
                          dtype=torch.int32, device=dev)
    totals = torch.empty(nd, dtype=torch.int32, device=dev)
    if hash_mix and nbits_eff != nbits:
        raise ValueError("hash partitioning requires nbits >= 4")
    s = _stream()
    m.radix_hist(keys.data_ptr(), n, shift, nbits_eff, hist.data_ptr(), s,
                 int(hash_mix))
    m.radix_scan(hist.data_ptr(), n, nbits_eff, totals.data_ptr(),
                 scan_ws.data_ptr(), s)
    keys_out = vals_out = None
    if key_dst is None:
        counts64 = totals.to(torch.int64)
        bases = torch.cumsum(counts64, 0) - counts64
        keys_out = torch.empty_like(keys)
        key_dst = keys_out.data_ptr() + bases * 8   # int64 device addresses
        if vals is not None:
            vals_out = torch.empty_like(vals)
            val_dst = vals_out.data_ptr() + bases * 8
        else:
            val_dst = torch.zeros(nd, dtype=torch.int64, device=dev)
    m.radix_scatter(keys.data_ptr(),
                    vals.data_ptr() if vals is not None else 0,
                    n, shift, nbits_eff, hist.data_ptr(),
                    key_dst.data_ptr(), val_dst.data_ptr(), s, int(hash_mix))
    return totals[:1 << nbits], keys_out, vals_out


def partition_aos(keys: torch.Tensor, vals: torch.Tensor, nbits: int,
                  shift: Optional[int] = None,
                  hash_mix: bool = False):
    """Partition SoA (keys, vals) into ONE interleaved AoS buffer with
    buckets laid out contiguously — the stage-mode (RCCL alltoallv) send
    buffer in a single scatter pass.

    Returns (counts int32[2^nbits] device, pairs int64[2n] device)."""
    m = load()
    n = keys.numel()
    if shift is None:
        shift = 64 - nbits
    nbits_eff = max(nbits, 4)
    nd = 1 << nbits_eff
    dev = keys.device
    hist = torch.empty(m.radix_hist_bytes(n, nbits_eff) // 4,
                       dtype=torch.int32, device=dev)
    scan_ws = torch.empty(m.radix_scan_ws_bytes(n, nbits_eff) // 4,
                          dtype=torch.int32, device=dev)
    totals = torch.empty(nd, dtype=torch.int32, device=dev)
    s = _stream()
    m.radix_hist(keys.data_ptr(), n, shift, nbits_eff, hist.data_ptr(), s,
                 int(hash_mix))
    m.radix_scan(hist.data_ptr(), n, nbits_eff, totals.data_ptr(),
                 scan_ws.data_ptr(), s)
    pairs = torch.empty(2 * n, dtype=torch.int64, device=dev)
    counts64 = totals.to(torch.int64)
    bases = torch.cumsum(counts64, 0) - counts64
    key_dst = pairs.data_ptr() + bases * 16
    val_dst = key_dst + 8
    m.radix_scatter(keys.data_ptr(), vals.data_ptr(), n, shift, nbits_eff,
                    hist.data_ptr(), key_dst.data_ptr(), val_dst.data_ptr(),
                    s, int(hash_mix), 1)
    return totals[:1 << nbits], pairs


def sort_pairs_aos(pairs: torch.Tensor, start_bit: int = 0,
                   end_bit: int = 64,
                   digit_bits: Optional[int] = None,
                   tmp: Optional[torch.Tensor] = None,
                   ws: Optional[torch.Tensor] = None) -> torch.Tensor:
    """Sort interleaved (key u64, val u64) 16-byte records by key bits
    [start_bit, end_bit). ``pairs`` is an int64 tensor of 2n elements.
    One dwordx4 load + one dwordx4 store per record per pass, and digit
    write bursts twice as long as the SoA path (measured 2x scattered
    write bandwidth at 256 B vs 128 B — profiles/r01_kernel_profile.md).
    """
    m = load()
    n = pairs.numel() // 2
    if n == 0:
        return pairs
    bits = min(end_bit, 64) - start_bit
    if digit_bits is None:
        # measured: 7-bit digits (512 B bursts) do NOT beat 8-bit — the
        # write-only burst model overestimates; pass floor is elsewhere
        digit_bits = 8
    passes = -(-bits // digit_bits)
    if tmp is None:
        tmp = torch.empty_like(pairs)
    else:
        assert tmp.numel() >= pairs.numel(), "tmp too small"
        tmp = tmp[:pairs.numel()]
    need_ws = m.onesweep_workspace_bytes(n, passes)
    if ws is None:
        ws = torch.empty(need_ws, dtype=torch.uint8, device=pairs.device)
    else:
        assert ws.numel() >= need_ws, "ws too small"
    fn = (m.onesweep_sort_aos7_u64 if digit_bits == 7
          else m.onesweep_sort_aos_u64)
    res = fn(pairs.data_ptr(), tmp.data_ptr(), n, start_bit,
             start_bit + bits, ws.data_ptr(), _stream())
    return pairs if res == 0 else tmp


def extract_pairs(recs: torch.Tensor, rec_bytes: int, key_bytes: int = 8,
                  pairs: Optional[torch.Tensor] = None) -> torch.Tensor:
    """records -> interleaved (key-prefix u64, aux u64) pairs, where
    aux = key_lo16 << 48 | record_index (kernels.hip extract_pairs)."""
    m = load()
    n = recs.numel() // rec_bytes
    if pairs is None:
        pairs = torch.empty(2 * n, dtype=torch.int64, device=recs.device)
    else:
        assert pairs.numel() >= 2 * n
        pairs = pairs[:2 * n]
    m.extract_pairs(recs.data_ptr(), n, rec_bytes, key_bytes,
                    pairs.data_ptr(), _stream())
    return pairs


def sort_records(recs: torch.Tensor, rec_bytes: int, key_bytes: int = 8,
                 end_bit: int = 64, out: Optional[torch.Tensor] = None,
                 pairs: Optional[torch.Tensor] = None,
                 tmp: Optional[torch.Tensor] = None,
                 ws: Optional[torch.Tensor] = None,
                 fuse: bool = True,
                 pairs_filled: bool = False) -> torch.Tensor:
    """Sort W-byte AoS records by their 80-bit key: u64 LE prefix at
    offset 0 (bits [0, end_bit) significant — callers whose partitions
    share top prefix bits pass end_bit = 64 - shared) then, for
    key_bytes == 10, a u16 LE low word at offset 8.

    Only 16-byte (prefix, aux) pairs flow through the radix passes; one
    gather pass then moves each record once — pair passes cost 16% of
    full-record passes at W = 100 (kernels.hip 'wide-record machinery').
    Returns the sorted records (a fresh tensor, or ``out``).
    """
    m = load()
    n = recs.numel() // rec_bytes
    if n == 0:
        return recs
    dev = recs.device
    if pairs_filled:
        # caller already extracted (e.g. incrementally per fetched chunk,
        # overlapping the fetch phase)
        assert pairs is not None and pairs.numel() >= 2 * n
        prs = pairs[:2 * n]
    else:
        prs = extract_pairs(recs, rec_bytes, key_bytes, pairs)
    if tmp is None:
        tmp = torch.empty_like(prs)
    else:
        assert tmp.numel() >= 2 * n
        tmp = tmp[:2 * n]
    passes = -(-end_bit // 8) + (2 if key_bytes > 8 else 0)
    need_ws = m.onesweep_workspace_bytes(n, passes)
    if ws is None:
        ws = torch.empty(need_ws, dtype=torch.uint8, device=dev)
    else:
        assert ws.numel() >= need_ws
    cur, other = prs, tmp
    if key_bytes > 8:
        # LSD over the 80-bit key: 16 aux bits first, then the prefix
        r = m.onesweep_sort_aos_word_u64(cur.data_ptr(), other.data_ptr(),
                                         n, 48, 64, ws.data_ptr(),
                                         _stream(), 1)
        if r == 1:
            cur, other = other, cur
    if out is None:
        out = torch.empty_like(recs)
    else:
        assert out.numel() >= recs.numel()
        out = out[:recs.numel()]
    if fuse:
        # the FINAL prefix pass writes whole records (kernels.hip
        # FUSE_GATHER): the last pair writeout + the separate gather's
        # pair re-read disappear. -1 = fusion not applicable under the
        # current ablation globals -> fall through to the plain path
        res = m.onesweep_sort_aos_fused_u64(
            cur.data_ptr(), other.data_ptr(), n, 0, min(end_bit, 64),
            ws.data_ptr(), _stream(), 0, recs.data_ptr(),
            out.data_ptr(), rec_bytes)
        if res != -1:
            return out
    r = m.onesweep_sort_aos_word_u64(cur.data_ptr(), other.data_ptr(), n,
                                     0, min(end_bit, 64), ws.data_ptr(),
                                     _stream(), 0)
    if r == 1:
        cur, other = other, cur
    m.gather_records(recs.data_ptr(), cur.data_ptr(), n, rec_bytes, 0,
                     out.data_ptr(), 0, 0, 0, 0, 0, 0, _stream())
    return out


def partition_records(recs: torch.Tensor, rec_bytes: int,
                      key_bytes: int = 8, nbits: int = 8,
                      shift: Optional[int] = None, func: int = 0,
                      nparts: int = 0):
    """Group W-byte records into 2^nbits contiguous buckets (digit from
    the key prefix) — the stage-mode (RCCL alltoall) send-buffer builder
    for wide records. Returns (counts int64 numpy, grouped records)."""
    m = load()
    n = recs.numel() // rec_bytes
    if shift is None:
        shift = 64 - nbits
    nbits_eff = max(nbits, 4)
    nd = 1 << nbits_eff
    dev = recs.device
    s = _stream()
    pairs = extract_pairs(recs, rec_bytes, key_bytes)
    hist = torch.empty(m.radix_hist_bytes(n, nbits_eff) // 4,
                       dtype=torch.int32, device=dev)
    scan_ws = torch.empty(m.radix_scan_ws_bytes(n, nbits_eff) // 4,
                          dtype=torch.int32, device=dev)
    totals = torch.empty(nd, dtype=torch.int32, device=dev)
    m.radix_hist(pairs.data_ptr(), n, shift, nbits_eff, hist.data_ptr(), s,
                 func, 2, nparts)
    m.radix_scan(hist.data_ptr(), n, nbits_eff, totals.data_ptr(),
                 scan_ws.data_ptr(), s)
    import numpy as np
    counts = totals.cpu().numpy().astype(np.int64)  # syncs the stream
    starts = np.zeros(nd, dtype=np.int64)
    np.cumsum(counts[:-1], out=starts[1:])
    pairs_out = torch.empty_like(pairs)
    bases_t = torch.from_numpy(starts).to(dev)
    kd = pairs_out.data_ptr() + bases_t * 16
    vd = kd + 8
    m.radix_scatter(pairs.data_ptr(), pairs.data_ptr() + 8, n, shift,
                    nbits_eff, hist.data_ptr(), kd.data_ptr(),
                    vd.data_ptr(), s, func, 1, 2, nparts)
    out = torch.empty_like(recs)
    dstart_t = torch.from_numpy(starts.astype(np.uint32)
                                .view(np.int32)).to(dev)
    dst_addr_t = out.data_ptr() + bases_t * rec_bytes
    m.gather_records(recs.data_ptr(), pairs_out.data_ptr(), n, rec_bytes,
                     1, 0, dst_addr_t.data_ptr(), dstart_t.data_ptr(),
                     shift, nd - 1, func, nparts, s)
    return counts[:1 << nbits], out


def sort_pairs(keys: torch.Tensor, vals: Optional[torch.Tensor] = None,
               start_bit: int = 0, end_bit: int = 64,
               onesweep: Optional[bool] = None
               ) -> Tuple[torch.Tensor, Optional[torch.Tensor]]:
    """LSD radix sort of u64 keys (and optional u64 payload) on device.

    Sorts by bits [start_bit, end_bit) — callers that range-partitioned by
    the top bits pass end_bit = 64 - log2(R) and save whole passes.
    Uses the decoupled-lookback onesweep path (one kernel per pass) when
    n < 2^30; the 3-kernel hist/scan/scatter path otherwise.
    """
    m = load()
    n = keys.numel()
    if n == 0:
        return keys, vals
    dev = keys.device
    tmp_k = torch.empty_like(keys)
    tmp_v = torch.empty_like(vals) if vals is not None else None
    end_bit = start_bit + ((end_bit - start_bit + 7) // 8) * 8  # whole digits
    end_bit = min(end_bit, 64)
    passes = (end_bit - start_bit) // 8
    if onesweep is None:
        onesweep = n < (1 << 30)
    if onesweep:
        ws = torch.empty(m.onesweep_workspace_bytes(n, passes),
                         dtype=torch.uint8, device=dev)
        res = m.onesweep_sort_pairs_u64(
            keys.data_ptr(), vals.data_ptr() if vals is not None else 0,
            tmp_k.data_ptr(), tmp_v.data_ptr() if tmp_v is not None else 0,
            n, start_bit, end_bit, ws.data_ptr(), _stream())
    else:
        ws = torch.empty(m.sort_workspace_bytes(n), dtype=torch.uint8,
                         device=dev)
        res = m.sort_pairs_u64(
            keys.data_ptr(), vals.data_ptr() if vals is not None else 0,
            tmp_k.data_ptr(), tmp_v.data_ptr() if tmp_v is not None else 0,
            n, start_bit, end_bit, ws.data_ptr(), _stream())
    if res == 0:
        return keys, vals
    return tmp_k, tmp_v


class AosSorter:
    """Persistent-buffer AoS sorter: workspace and ping-pong buffer are
    allocated once, so a sort issues ONLY kernel launches and stream
    memsets. (Intended to make the launch sequence hipGraph-capturable;
    torch.cuda.graph capture of the extension's launches SIGABRTs on this
    ROCm/torch build — see ROADMAP.md — so capture is not wired up.)"""

    def __init__(self, n: int, device="cuda", start_bit: int = 0,
                 end_bit: int = 64):
        m = load()
        self.n = n
        self.start_bit = start_bit
        self.end_bit = min(start_bit + ((end_bit - start_bit + 7) // 8) * 8, 64)
        passes = (self.end_bit - start_bit) // 8
        self.tmp = torch.empty(2 * n, dtype=torch.int64, device=device)
        self.ws = torch.empty(m.onesweep_workspace_bytes(n, passes),
                              dtype=torch.uint8, device=device)
        self._m = m

    def sort_(self, pairs: torch.Tensor) -> torch.Tensor:
        assert pairs.numel() == 2 * self.n
        res = self._m.onesweep_sort_aos_u64(
            pairs.data_ptr(), self.tmp.data_ptr(), self.n, self.start_bit,
            self.end_bit, self.ws.data_ptr(), _stream())
        return pairs if res == 0 else self.tmp
