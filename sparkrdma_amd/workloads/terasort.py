"""TeraSort — the reference's headline workload (README.md:7-17).

Synthetic records: canonical 100-byte records (10 B key = u64 LE prefix +
u16 LE low, 90 B payload) on the wide path, or (u64 key, u64 payload)
16-byte pairs. One *step* is a complete sort job over the fixed
per-executor dataset:

  map:    radix-partition the local shard by the top log2(R) key bits into
          R global partitions, serialized straight into HBM blocks
          (GPU) or shm blocks (CPU)
  shuffle: every executor one-sidedly fetches its partition range
           (xGMI peer copies / shm preads), or RCCL all-to-all in
           stage mode
  reduce: LSD radix sort of each partition over the REMAINING low bits
          (the top bits are constant within a partition — the global
          partitioning is the MSD pass)

Output invariant (validated): concatenated partitions are globally sorted
and are a permutation of the input.
"""

from __future__ import annotations

import logging
import time
from dataclasses import dataclass
from typing import Optional

import numpy as np

from ..engine import Engine
from ..partitioner import RangePartitioner
from ..writer import unpack_partition_segment

log = logging.getLogger(__name__)


@dataclass
class TeraSortResult:
    seconds: float
    records: int
    bytes_sorted: int
    write_s: float
    fetch_s: float
    sort_s: float
    remote_bytes: int


class TeraSort:
    """record_bytes = 16: (u64 key, u64 payload) AoS — the r01 fast shape.
    record_bytes = 100 (canonical): 10 B key (u64 LE prefix + u16 LE low)
    + 90 B payload, the reference benchmark's true record
    (README.md:7-17); GPU-only (wide-record kernels)."""

    RECORD_BYTES = 16

    def __init__(self, engine: Engine, records_per_executor: int,
                 partitions_per_executor: int = 64,
                 device: str = "cpu", mode: str = "framework",
                 validate: bool = False, seed: int = 0,
                 record_bytes: int = 16):
        self.engine = engine
        self.n = records_per_executor
        self.device = device
        self.mode = mode
        self.validate = validate
        self.RECORD_BYTES = record_bytes
        self.wide = record_bytes != 16
        if self.wide and device != "cuda":
            raise ValueError("wide records need device=cuda")
        W = engine.world_size
        # R must be pow2 for the GPU top-bits partitioner
        R = W * partitions_per_executor
        if R & (R - 1):
            raise ValueError("world_size * partitions_per_executor must be pow2")
        self.R = R
        self.ppe = partitions_per_executor
        self.part = RangePartitioner.uniform(R)
        self.low_bits = 64 - (R - 1).bit_length()
        rank = engine.rank
        if self.wide:
            import torch
            g = torch.Generator(device="cuda").manual_seed(seed * 1000 + rank)
            Wb = record_bytes
            # record: [key u64 LE | keylo u16 | payload]; payload's first
            # 8 bytes mirror the prefix for integrity validation
            recs = torch.randint(-128, 128, (self.n, Wb), dtype=torch.int8,
                                 device="cuda", generator=g)
            prefix = recs[:, :8]
            recs[:, 10:18] = prefix   # integrity mirror
            self.recs = recs.view(torch.uint8).reshape(-1)
            self.keys = self.vals = None
        elif device == "cuda":
            import torch
            g = torch.Generator(device="cuda").manual_seed(seed * 1000 + rank)
            self.keys = torch.randint(-2**63, 2**63 - 1, (self.n,),
                                      dtype=torch.int64, device="cuda",
                                      generator=g)
            self.vals = self.keys.clone()  # payload := key bytes (integrity)
        else:
            rng = np.random.default_rng(seed * 1000 + rank)
            self.keys = rng.integers(0, 2**64, self.n, dtype=np.uint64)
            self.vals = self.keys.view(np.uint8).reshape(-1, 8).copy()

    # ------------------------------------------------------------------

    def run_step(self) -> TeraSortResult:
        if self.mode == "rccl":
            return self._step_rccl()
        if self.mode == "shuffleread":
            return self._step_shuffleread()
        return self._step_framework()

    def _step_shuffleread(self) -> TeraSortResult:
        """Pure one-sided read bandwidth: the map output is written once
        (first step); every step then re-fetches this rank's partition
        range over xGMI/shm — BASELINE's 'shuffle-read GB/s at 1/2/4/8'
        measured directly, no sort in the timed path."""
        eng = self.engine
        rank = eng.rank
        t0 = time.perf_counter()
        if getattr(self, "_sr_handle", None) is None:
            handle = eng.register_shuffle(eng.world_size, self.R)
            w = eng.manager.get_writer(handle, rank)
            if self.wide:
                w.write_device_records(self.recs, self.RECORD_BYTES, 10)
            elif self.device == "cuda":
                w.write_device_batch(self.keys, self.vals)
            else:
                w.write_batch(self.keys, self.vals)
            w.stop(True, partitioner=self.part)
            eng.barrier()
            self._sr_handle = handle
        t_write = time.perf_counter()
        lo, hi = rank * self.ppe, (rank + 1) * self.ppe - 1
        arena = self._arenas(1)[0]
        reader = eng.manager.get_reader(self._sr_handle, lo, hi, arena=arena)
        n_blocks = sum(1 for _ in reader)
        if self.device == "cuda":
            import torch
            torch.cuda.synchronize()
        t_fetch = time.perf_counter()
        eng.barrier()
        dt = time.perf_counter() - t0
        fetched = (reader.metrics.remote_bytes_read
                   + reader.metrics.local_bytes_read)
        return TeraSortResult(
            seconds=dt, records=self.n, bytes_sorted=fetched,
            write_s=t_write - t0, fetch_s=t_fetch - t_write, sort_s=0.0,
            remote_bytes=reader.metrics.remote_bytes_read)

    def _step_framework(self) -> TeraSortResult:
        eng = self.engine
        rank = eng.rank
        t0 = time.perf_counter()
        handle = eng.register_shuffle(eng.world_size, self.R)
        w = eng.manager.get_writer(handle, rank)
        if self.wide:
            w.write_device_records(self.recs, self.RECORD_BYTES, 10)
        elif self.device == "cuda":
            w.write_device_batch(self.keys, self.vals)
        else:
            w.write_batch(self.keys, self.vals)
        w.stop(True, partitioner=self.part)
        t_write = time.perf_counter()
        eng.barrier()
        lo, hi = rank * self.ppe, (rank + 1) * self.ppe - 1
        # pipelined reduce: H chunk readers start fetching concurrently at
        # construction; chunk h sorts while later chunks' one-sided copies
        # are still in flight — overlapping remote xGMI fetches with the
        # radix sort. Chunking GROWS the shared top bits (per-chunk pid
        # ranges are pow2-aligned) so it never adds a radix pass; at
        # world=1 the local D2D fetch is already stream-overlapped and
        # H=1 measured fastest (253.7 vs 257.8 GB/s — r02 A/B).
        import os as _os
        H = int(_os.environ.get("TERASORT_H", 0)) or \
            (1 if eng.world_size == 1 else min(4, self.ppe))
        # ppe is pow2, so per = ppe/H is exact
        per = self.ppe // H
        spans = [(lo + h * per, lo + (h + 1) * per - 1) for h in range(H)]
        arenas = self._arenas(H)
        readers = [eng.manager.get_reader(handle, a, b, arena=ar)
                   for (a, b), ar in zip(spans, arenas)]
        t_fetch_total = 0.0
        t_sort_total = 0.0
        outs = []
        remote = 0
        chunk_shared_bits = (self.R // per - 1).bit_length()
        for (chunk_lo, _chunk_hi), reader in zip(spans, readers):
            tf = time.perf_counter()
            if self.wide:
                # drive the iterator OURSELVES: each fetched chunk's pair
                # extraction launches as the chunk lands, overlapping the
                # remaining fetches (collect_partitions is not needed —
                # the arena holds everything)
                pre_extracted = self._extract_while_fetching(reader)
                parts = None
            else:
                parts = reader.collect_partitions()
            ts_ = time.perf_counter()
            arena = getattr(reader.fetcher, "arena", None)
            if self.wide and arena is not None:
                import torch
                from ..ops.radix import sort_records
                nrec = arena.numel() // self.RECORD_BYTES
                out = sort_records(
                    arena, self.RECORD_BYTES, key_bytes=10,
                    end_bit=64 - chunk_shared_bits,
                    out=self._rec_out(arena.numel()),
                    pairs=self._sort_tmp(2 * nrec, "_pairs_cache"),
                    tmp=self._sort_tmp(2 * nrec, "_tmp_cache"),
                    ws=self._sort_ws(),
                    pairs_filled=pre_extracted)
                torch.cuda.synchronize()
                outs.append(out)
            elif self.device == "cuda" and arena is not None:
                # fetches landed pre-placed in one device buffer: sort it
                # directly (no concat pass); tmp/ws persist across steps
                # so a 40 GB step performs no large allocations at all
                import torch
                from ..ops.radix import sort_pairs_aos
                pairs = sort_pairs_aos(arena.view(torch.int64), 0,
                                       64 - chunk_shared_bits,
                                       tmp=self._sort_tmp(arena.numel() // 8),
                                       ws=self._sort_ws())
                torch.cuda.synchronize()
                outs.append((pairs[0::2], pairs[1::2]))
            else:
                outs.append(self._reduce(parts, chunk_shared_bits))
            t_fetch_total += ts_ - tf
            t_sort_total += time.perf_counter() - ts_
            remote += reader.metrics.remote_bytes_read
            if self.validate:
                # validate THIS chunk now: the sort tmp/out caches are
                # shared across chunks, so chunk h+1 overwrites h's view
                self._validate(outs[-1], chunk_lo, per)
        eng.unregister_shuffle(handle)
        dt = time.perf_counter() - t0
        return TeraSortResult(
            seconds=dt, records=self.n,
            bytes_sorted=self.n * self.RECORD_BYTES,
            write_s=t_write - t0, fetch_s=t_fetch_total,
            sort_s=t_sort_total, remote_bytes=remote)

    def _arenas(self, H: int):
        """Reusable per-chunk fetch arenas (slack for partition skew)."""
        if self.device != "cuda":
            return [None] * H
        import torch
        cap = int(self.n * self.RECORD_BYTES // H * 1.25) + (64 << 10)
        cur = getattr(self, "_arena_cache", None)
        if cur is None or len(cur) != H or cur[0].numel() < cap:
            self._arena_cache = [torch.empty(cap, dtype=torch.uint8,
                                             device="cuda")
                                 for _ in range(H)]
        return self._arena_cache

    def _extract_while_fetching(self, reader) -> bool:
        """Iterate the fetcher, launching the (prefix, aux) pair
        extraction for each CONTIGUOUS landed run — the extract overlaps
        the in-flight fetches instead of re-reading the full arena
        afterwards. Returns True when every byte was covered (else the
        caller extracts in one pass — growth/non-tensor fallback)."""
        import torch
        from ..ops import load
        hs = load()
        W = self.RECORD_BYTES
        f = reader.fetcher
        buf = f._arena_buf
        if buf is None:
            for _ in f:
                pass
            return False
        base = buf.data_ptr()
        cap = buf.numel()
        pairs_buf = self._sort_tmp(2 * (cap // W) + 16, "_pairs_cache")
        stream = torch.cuda.current_stream().cuda_stream
        ok = True
        covered = 0
        run_off = run_len = 0

        def flush():
            nonlocal ok, covered
            if not run_len:
                return
            if run_off % W or run_len % W:
                ok = False
                return
            hs.extract_pairs(base + run_off, run_len // W, W, 10,
                             pairs_buf.data_ptr() + (run_off // W) * 16,
                             stream, idx_base=run_off // W)
            covered += run_len

        for _ref, view in f:
            if (not isinstance(view, torch.Tensor)
                    or f._arena_buf is not buf):
                ok = False       # grew / non-arena chunk: full re-extract
                continue
            boff = view.data_ptr() - base
            nb = view.numel()
            if not (0 <= boff and boff + nb <= cap):
                ok = False
                continue
            if boff == run_off + run_len:
                run_len += nb
            else:
                flush()
                run_off, run_len = boff, nb
        flush()
        arena = f.arena
        return bool(ok and arena is not None
                    and covered == arena.numel())

    def _sort_tmp(self, n_i64: int, attr: str = "_tmp_cache"):
        import torch
        cur = getattr(self, attr, None)
        if cur is None or cur.numel() < n_i64:
            setattr(self, attr, torch.empty(int(n_i64 * 1.05) + 1024,
                                            dtype=torch.int64, device="cuda"))
        return getattr(self, attr)

    def _rec_out(self, nbytes: int):
        import torch
        cur = getattr(self, "_rec_out_cache", None)
        if cur is None or cur.numel() < nbytes:
            self._rec_out_cache = torch.empty(int(nbytes * 1.05) + 4096,
                                              dtype=torch.uint8,
                                              device="cuda")
        return self._rec_out_cache

    def _sort_ws(self):
        from ..ops import load
        import torch
        cur = getattr(self, "_ws_cache", None)
        need = load().onesweep_workspace_bytes(
            int(self.n * 1.25) + 4096, 10)
        if cur is None or cur.numel() < need:
            self._ws_cache = torch.empty(need, dtype=torch.uint8,
                                         device="cuda")
        return self._ws_cache

    def _reduce(self, parts: dict, shared_bits: Optional[int] = None):
        """One batched sort per fetched chunk: all partitions of a chunk
        share their top `shared_bits` key bits (the chunk's partition
        prefix), so sorting the concatenated fetches over the remaining
        low bits yields the chunk's fully sorted output in a single kernel
        sequence — avoiding per-partition launch/workspace overhead
        (measured: R=64 beat R=256 at 8 GB despite one extra radix pass)."""
        wbits = (shared_bits if shared_bits is not None
                 else (self.engine.world_size - 1).bit_length())
        if self.device == "cuda":
            import torch
            from ..ops.radix import sort_pairs_aos
            from ..utils import as_device_i64
            # fetched chunks are AoS (key,val) records: concat directly
            # (host-spilled chunks upload transparently)
            ts = [as_device_i64(c) for chunks in parts.values()
                  for c in chunks]
            if not ts:
                return None, None
            pairs = torch.cat(ts) if len(ts) > 1 else ts[0].contiguous()
            pairs = sort_pairs_aos(pairs, 0, 64 - wbits)
            torch.cuda.synchronize()
            return pairs[0::2], pairs[1::2]  # views; no copy until needed
        ks, vs = [], []
        for p, chunks in parts.items():
            for c in chunks:
                k, v = unpack_partition_segment(c, 8)
                ks.append(np.array(k))
                vs.append(np.array(v))
        if not ks:
            return None, None
        k = np.concatenate(ks)
        v = np.concatenate(vs)
        order = np.argsort(k, kind="stable")
        return k[order], v[order]

    def _validate(self, sorted_out, lo: int, span: Optional[int] = None) -> None:
        if self.wide:
            return self._validate_wide(sorted_out, lo, span)
        k, v = sorted_out
        if k is None:
            return
        if self.device == "cuda":
            ku = k.cpu().numpy().view(np.uint64)
            vu = v.cpu().numpy().view(np.uint64)
        else:
            ku, vu = k, v.view(np.uint64).reshape(-1)
        assert np.all(ku[1:] >= ku[:-1]), "chunk output not sorted"
        assert np.array_equal(ku, vu.reshape(-1)), "payload corrupted"
        pids = self.part.partition_ids(ku)
        hi = lo + (span or self.ppe) - 1
        assert np.all((pids >= lo) & (pids <= hi)), "foreign keys in range"
        log.info("validated %d records in partitions [%d, %d]", len(ku), lo, hi)

    def _validate_wide(self, out, lo: int, span: Optional[int] = None) -> None:
        """Wide records: 80-bit key order (prefix, then u16 low bits),
        payload integrity via the mirrored prefix, partition membership."""
        W = self.RECORD_BYTES
        arr = out.cpu().numpy().reshape(-1, W)
        prefix = arr[:, :8].copy().view("<u8").ravel()
        lo16 = arr[:, 8:10].copy().view("<u2").ravel()
        mirror = arr[:, 10:18].copy().view("<u8").ravel()
        assert np.all(prefix[1:] >= prefix[:-1]), "prefixes not sorted"
        ties = prefix[1:] == prefix[:-1]
        assert np.all(lo16[1:][ties] >= lo16[:-1][ties]), \
            "low key bits not sorted within prefix ties"
        assert np.array_equal(mirror, prefix), "payload corrupted"
        pids = self.part.partition_ids(prefix)
        hi = lo + (span or self.ppe) - 1
        assert np.all((pids >= lo) & (pids <= hi)), "foreign keys in range"
        log.info("validated %d wide records in partitions [%d, %d]",
                 len(prefix), lo, hi)

    # ------------------------------------------------------------------

    def _step_rccl(self) -> TeraSortResult:
        """Stage-mode shuffle: ONE RCCL all_to_all_single over xGMI of
        AoS records (SURVEY §7.1 'collective-form option') — the partition
        pass scatters straight into the send buffer. Wide records ship as
        uint8 with byte splits (partition_records builds the grouped send
        buffer; the reduce side is the same pair-indirection sort)."""
        import torch
        import torch.distributed as dist
        from ..ops.radix import partition_aos, sort_pairs_aos
        eng = self.engine
        W = eng.world_size
        t0 = time.perf_counter()
        wbits = (W - 1).bit_length()
        if self.wide:
            return self._step_rccl_wide(W, wbits, t0)
        if W == 1:
            pairs = torch.empty(2 * self.n, dtype=torch.int64, device="cuda")
            pairs[0::2] = self.keys
            pairs[1::2] = self.vals
            pairs = sort_pairs_aos(pairs, 0, 64)
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            if self.validate:
                ku = pairs[0::2].cpu().numpy().view(np.uint64)
                assert np.all(ku[1:] >= ku[:-1])
            return TeraSortResult(dt, self.n, self.n * 16, 0, 0, dt, 0)
        counts, send_pairs = partition_aos(self.keys, self.vals, wbits,
                                           shift=64 - wbits)
        in_splits = counts.to(torch.int64) * 2   # i64 elements per record: 2
        out_splits = torch.empty_like(in_splits)
        dist.all_to_all_single(out_splits, in_splits)
        in_l = in_splits.cpu().tolist()
        out_l = out_splits.cpu().tolist()
        recv = torch.empty(sum(out_l), dtype=torch.int64, device="cuda")
        dist.all_to_all_single(recv, send_pairs, out_l, in_l)
        t_fetch = time.perf_counter()
        pairs = sort_pairs_aos(recv, 0, 64 - wbits)
        torch.cuda.synchronize()
        t_sort = time.perf_counter()
        if self.validate:
            ku = pairs[0::2].cpu().numpy().view(np.uint64)
            assert np.all(ku[1:] >= ku[:-1])
        dt = time.perf_counter() - t0
        return TeraSortResult(dt, self.n, self.n * 16,
                              t_fetch - t0, 0, t_sort - t_fetch,
                              int(sum(out_l)) * 8)

    def _step_rccl_wide(self, W: int, wbits: int, t0: float) -> TeraSortResult:
        import torch
        import torch.distributed as dist
        from ..ops.radix import partition_records, sort_records
        Wb = self.RECORD_BYTES
        if W == 1:
            out = sort_records(self.recs, Wb, key_bytes=10)
            torch.cuda.synchronize()
            dt = time.perf_counter() - t0
            if self.validate:
                self._validate_wide(out, 0, self.R)
            return TeraSortResult(dt, self.n, self.n * Wb, 0, 0, dt, 0)
        counts, grouped = partition_records(self.recs, Wb, key_bytes=10,
                                            nbits=wbits, shift=64 - wbits)
        in_t = torch.from_numpy(counts * Wb).cuda()        # bytes per rank
        out_t = torch.empty_like(in_t)
        dist.all_to_all_single(out_t, in_t)
        in_l = (counts * Wb).tolist()
        out_l = out_t.cpu().tolist()
        recv = torch.empty(int(sum(out_l)), dtype=torch.uint8,
                           device="cuda")
        dist.all_to_all_single(recv, grouped, out_l, in_l)
        t_fetch = time.perf_counter()
        out = sort_records(recv, Wb, key_bytes=10, end_bit=64 - wbits)
        torch.cuda.synchronize()
        t_sort = time.perf_counter()
        if self.validate:
            # rank r holds pids with top wbits == r == [r*ppe, (r+1)*ppe)
            self._validate_wide(out, self.engine.rank * self.ppe, self.ppe)
        dt = time.perf_counter() - t0
        return TeraSortResult(dt, self.n, self.n * Wb,
                              t_fetch - t0, 0, t_sort - t_fetch,
                              int(sum(out_l)))
