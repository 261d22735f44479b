"""Shuffle write path.

Re-design of the reference write path (RdmaWrapperShuffleWriter.scala +
RdmaMappedFile.java). The reference delegates record partitioning to
Spark's CPU sort-shuffle writers, then mmaps + registers the resulting
data file and fills a per-partition (addr, len, key) table
(RdmaMappedFile.java:113-157). The MI355X design removes the file from the
fast path entirely: partitioned bytes are laid out directly in pool blocks
(host shm now; HBM slabs via the GPU writer in ops/), the table is filled
from the layout, and publishing is a one-sided 12-byte write.

Block chunking keeps the reference policy (RdmaMappedFile.java:113-157):
partitions are packed greedily into ~shuffle_write_block_size blocks, split
only at partition boundaries, so every partition is one contiguous
(addr, len) range.

Record model (fixed-width path): a partition segment is AoS —
``[key u64 | value bytes] × n`` interleaved records (doubles the GPU
scatter's per-digit write bursts and lets the reduce side consume fetched
chunks without splitting); n is inferable from the segment length. The
bytes path (arbitrary pickled records) concatenates pickle frames.
"""

from __future__ import annotations

import pickle
import time
from typing import List, Optional

import numpy as np

from .manager import ShuffleHandle, ShuffleManager
from .map_output import make_key
from .stats import TaskMetrics

HEADER_W = 8  # bytes per key


def pack_partition_segment(keys: np.ndarray, values: Optional[np.ndarray]) -> bytes:
    """AoS record layout: [key u64 | value bytes] interleaved per record.

    Interleaving doubles the GPU scatter's per-digit write-burst length
    (the partition kernel writes the identical layout) and lets the
    reduce side consume fetched chunks without a split/concat step.
    """
    if values is None or values.size == 0:
        return keys.astype("<u8", copy=False).tobytes()
    n = len(keys)
    vw = values.shape[1]
    rec = np.empty((n, HEADER_W + vw), dtype=np.uint8)
    rec[:, :HEADER_W] = keys.astype("<u8", copy=False).view(np.uint8).reshape(n, 8)
    rec[:, HEADER_W:] = values
    return rec.tobytes()


def unpack_partition_segment(buf, value_width: int):
    if value_width == 0:
        return np.frombuffer(buf, dtype="<u8"), None
    rec_w = HEADER_W + value_width
    n = len(buf) // rec_w
    arr = np.frombuffer(buf, dtype=np.uint8)[:n * rec_w].reshape(n, rec_w)
    keys = arr[:, :HEADER_W].copy().view("<u8").reshape(n)
    values = arr[:, HEADER_W:]
    return keys, values


class ShuffleWriter:
    """One map task's writer: accumulate → partition → lay out → publish."""

    def __init__(self, manager: ShuffleManager, handle: ShuffleHandle, map_id: int):
        self.manager = manager
        self.handle = handle
        self.map_id = map_id
        self.metrics = TaskMetrics()
        self._key_batches: List[np.ndarray] = []
        self._value_batches: List[Optional[np.ndarray]] = []
        self._byte_records: Optional[List[List[bytes]]] = None
        self._stopped = False

    # -- fixed-width tensor path ---------------------------------------

    def write_batch(self, keys: np.ndarray, values: Optional[np.ndarray] = None) -> None:
        if values is not None and len(values) != len(keys):
            raise ValueError("keys/values length mismatch")
        self._key_batches.append(np.ascontiguousarray(keys, dtype=np.uint64))
        self._value_batches.append(
            None if values is None else np.ascontiguousarray(values, dtype=np.uint8))

    # -- GPU tensor path: CDNA4 partition kernel into HBM blocks --------

    def write_device_batch(self, keys, values=None) -> None:
        """keys/values: int64 CUDA tensors. Partitioned and serialized by
        the radix kernel straight into HBM pool blocks at stop()."""
        self._gpu_batches = getattr(self, "_gpu_batches", [])
        self._gpu_batches.append((keys, values))

    def write_device_records(self, records, record_bytes: int,
                             key_bytes: int = 8) -> None:
        """Wide-record GPU path (canonical TeraSort: 100-byte records =
        10 B key + 90 B value). ``records`` is a flat uint8 CUDA tensor of
        n*record_bytes; key layout: u64 LE prefix at offset 0 (+ u16 LE
        low bits at offset 8 when key_bytes == 10). The reference's
        contract is 'serve whatever bytes the writer produced'
        (RdmaMappedFile.java:113-157) — this is the GPU fast path for it
        (r01 handled only 16-byte records)."""
        if record_bytes % 4 or record_bytes < 8:
            raise ValueError("record_bytes must be a multiple of 4, >= 8")
        if key_bytes not in (8, 10):
            raise ValueError("key_bytes must be 8 or 10")
        self._gpu_records = getattr(self, "_gpu_records", [])
        if records.numel() % record_bytes:
            raise ValueError("records length not a record multiple")
        prev = getattr(self, "_record_shape", None)
        if prev is not None and prev != (record_bytes, key_bytes):
            raise ValueError("record shape changed between batches")
        self._record_shape = (record_bytes, key_bytes)
        self._gpu_records.append(records)

    # -- arbitrary-record path -----------------------------------------

    def write_records(self, records, partitioner=None) -> None:
        """records: iterable of (key, value) python objects. Partition by
        ``partitioner`` when given (a callable key->pid, or an object with
        ``partition_ids`` for integer keys); default is pickled-key crc32
        (stable across processes)."""
        R = self.handle.num_partitions
        if self._byte_records is None:
            self._byte_records = [[] for _ in range(R)]
        import zlib
        if partitioner is None:
            def pid_of(k):
                return zlib.crc32(pickle.dumps(k, protocol=4)) % R
        elif hasattr(partitioner, "partition_ids"):
            def pid_of(k):
                return int(partitioner.partition_ids(
                    np.array([k], dtype=np.uint64))[0])
        elif callable(partitioner):
            def pid_of(k):
                return int(partitioner(k)) % R
        else:
            raise TypeError(f"unsupported partitioner {partitioner!r}")
        for k, v in records:
            pid = pid_of(k)
            if not 0 <= pid < R:
                raise ValueError(f"partitioner returned {pid} for key {k!r}, "
                                 f"outside [0, {R})")
            self._byte_records[pid].append(pickle.dumps((k, v), protocol=4))
            self.metrics.records_written += 1

    # -- commit ---------------------------------------------------------

    def stop(self, success: bool, partitioner=None) -> None:
        if self._stopped:
            return
        self._stopped = True
        if not success:
            return
        t0 = time.perf_counter_ns()
        if getattr(self, "_gpu_records", None):
            self._commit_gpu_records(partitioner)
        elif getattr(self, "_gpu_batches", None):
            self._commit_gpu(partitioner)
        else:
            if self._byte_records is not None:
                segments = [b"".join(recs) for recs in self._byte_records]
            else:
                segments = self._partition_fixed(partitioner)
            self._commit_segments(segments)
        self.metrics.write_ns += time.perf_counter_ns() - t0
        lm = getattr(self.manager, "lifetime_metrics", None)
        if lm is not None:
            lm.merge(self.metrics)

    def _partition_fixed(self, partitioner) -> List[bytes]:
        R = self.handle.num_partitions
        if not self._key_batches:
            return [b""] * R
        keys = (self._key_batches[0] if len(self._key_batches) == 1
                else np.concatenate(self._key_batches))
        vals = None
        if self._value_batches and self._value_batches[0] is not None:
            vals = (self._value_batches[0] if len(self._value_batches) == 1
                    else np.concatenate(self._value_batches))
        pids = partitioner.partition_ids(keys)
        order = np.argsort(pids, kind="stable")
        keys_sorted = keys[order]
        vals_sorted = vals[order] if vals is not None else None
        counts = np.bincount(pids, minlength=R)
        ends = np.cumsum(counts)
        starts = ends - counts
        self.metrics.records_written += len(keys)
        return [pack_partition_segment(
                    keys_sorted[starts[p]:ends[p]],
                    vals_sorted[starts[p]:ends[p]] if vals_sorted is not None else None)
                for p in range(R)]

    def _gpu_part_params(self, partitioner):
        """(func, shift, nparts, nbits_eff) for the kernel-side partition
        function (partitioner.gpu_params()): pow2 R via bit extraction,
        arbitrary R via mulhi-range / hash-mod. One kernel pass covers
        nbits <= 12; larger R runs the two-level pid radix
        (_group_pairs_2level), capped at 2^24."""
        R = self.handle.num_partitions
        nbits = max((R - 1).bit_length(), 1)
        if nbits > 24:
            raise ValueError(
                f"GPU partitioner supports R <= 2^24, got {R}; "
                "use the CPU write path for larger partition counts")
        params = None
        if hasattr(partitioner, "gpu_params"):
            params = partitioner.gpu_params()
        elif hasattr(partitioner, "gpu_shift"):
            params = (0, partitioner.gpu_shift, 0)
        if params is None:
            raise ValueError(
                f"partitioner {partitioner!r} has no GPU dispatch "
                "(gpu_params() returned None) — use the CPU write path")
        func, shift, nparts = params
        return func, shift, nparts, min(max(nbits, 4), 12)

    def _commit_gpu(self, partitioner) -> None:
        """Map-side GPU write: one radix pass whose scatter writes each
        partition's [keys|vals] segment STRAIGHT into its final position in
        HBM pool blocks (the north-star replacement for the reference's
        CPU writer + mmap + register, RdmaMappedFile.java:113-189)."""
        import torch
        from .ops import load as ops_load
        mgr = self.manager
        if mgr.gpu is None:
            raise RuntimeError("GPU writer requires the GPU data plane")
        hs = ops_load()
        R = self.handle.num_partitions
        batches = self._gpu_batches
        keys = (batches[0][0] if len(batches) == 1
                else torch.cat([b[0] for b in batches]))
        vals = (batches[0][1] if len(batches) == 1 or batches[0][1] is None
                else torch.cat([b[1] for b in batches]))
        if (R - 1).bit_length() > 12:
            # > 4096 partitions: route through the wide-record two-level
            # pid radix (records = the interleaved 16-B AoS pairs, or the
            # bare keys when value-less)
            if vals is None:
                recs = keys.contiguous().view(torch.uint8).reshape(-1)
                self._record_shape = (8, 8)
            else:
                pr = torch.empty(2 * keys.numel(), dtype=torch.int64,
                                 device=keys.device)
                pr[0::2] = keys
                pr[1::2] = vals
                recs = pr.view(torch.uint8).reshape(-1)
                self._record_shape = (16, 8)
            self._gpu_records = [recs]
            return self._commit_gpu_records(partitioner)
        func, shift, nparts, nbits_eff = self._gpu_part_params(partitioner)
        has_val = vals is not None
        n = keys.numel()
        if n == 0:
            self._commit_empty()
            return
        dev = keys.device
        stream = torch.cuda.current_stream().cuda_stream
        nd = 1 << nbits_eff
        hist = torch.empty(hs.radix_hist_bytes(n, nbits_eff) // 4,
                           dtype=torch.int32, device=dev)
        scan_ws = torch.empty(hs.radix_scan_ws_bytes(n, nbits_eff) // 4,
                              dtype=torch.int32, device=dev)
        totals = torch.empty(nd, dtype=torch.int32, device=dev)
        hs.radix_hist(keys.data_ptr(), n, shift, nbits_eff, hist.data_ptr(),
                      stream, func, 1, nparts)
        hs.radix_scan(hist.data_ptr(), n, nbits_eff, totals.data_ptr(),
                      scan_ws.data_ptr(), stream)
        counts = totals.cpu().numpy().astype(np.int64)[:R]  # syncs the stream
        rec_w = HEADER_W + (8 if has_val else 0)
        seg_bytes = counts * rec_w

        # greedy chunking of partitions into HBM blocks (same policy as host)
        table, table_addr = mgr.alloc_table(R, self.handle.shuffle_id)
        write_block = mgr.conf.shuffle_write_block_size
        pool = mgr.gpu.pool
        blocks = []
        key_dst = np.zeros(nd, dtype=np.int64)
        val_dst = np.zeros(nd, dtype=np.int64)
        group, group_bytes = [], 0
        flushes = []
        for p in range(R):
            if group and group_bytes + seg_bytes[p] > write_block:
                flushes.append(group)
                group, group_bytes = [], 0
            group.append(p)
            group_bytes += int(seg_bytes[p])
        if group:
            flushes.append(group)
        meta_key = make_key(mgr.executor_id, 1)
        spill_groups = []   # (parts, total) that did not fit the HBM pool
        for parts in flushes:
            total = int(sum(seg_bytes[p] for p in parts))
            if total == 0:
                for p in parts:
                    table.put(p, 0, 0, meta_key)
                continue
            try:
                blk = pool.get(total)
            except MemoryError:
                # pool pressure: this group spills to a host block after
                # the scatter (reference keeps a disk file path for
                # overflow — SURVEY §7.1 RdmaMappedFile row)
                spill_groups.append((parts, total))
                continue
            base = mgr.gpu.local_base(blk.segment_id)
            key = make_key(mgr.executor_id, blk.segment_id)
            off = blk.offset
            for p in parts:
                nb = int(seg_bytes[p])
                table.put(p, off, nb, key)
                if nb:
                    # AoS segment: record r's key at +r*16, value at +r*16+8
                    key_dst[p] = base + off
                    val_dst[p] = base + off + (8 if has_val else 0)
                off += nb
                self.metrics.bytes_written += nb
            blocks.append(blk)
        spill_stage = None
        if spill_groups:
            # scatter spilled partitions into one transient device staging
            # tensor, then copy each group to a pooled host block
            stage_total = sum(t for _, t in spill_groups)
            spill_stage = torch.empty(stage_total, dtype=torch.uint8,
                                      device=dev)
            soff = 0
            for parts, total in spill_groups:
                for p in parts:
                    nb = int(seg_bytes[p])
                    if nb:
                        key_dst[p] = spill_stage.data_ptr() + soff
                        val_dst[p] = spill_stage.data_ptr() + soff + \
                            (8 if has_val else 0)
                    # table filled after host placement below
                    soff += nb
        kd = torch.from_numpy(key_dst).to(dev)
        vd = torch.from_numpy(val_dst).to(dev)
        hs.radix_scatter(keys.data_ptr(),
                         vals.data_ptr() if has_val else 0,
                         n, shift, nbits_eff, hist.data_ptr(),
                         kd.data_ptr(), vd.data_ptr(), stream, func,
                         1 if has_val else 0, 1, nparts)
        torch.cuda.synchronize()
        if spill_groups:
            soff = 0
            for parts, total in spill_groups:
                hblk = mgr.pool.get(total)     # host pool
                hseg = mgr.data_segment(hblk.segment_id)
                hkey = make_key(mgr.executor_id, hblk.segment_id)
                cpu_bytes = spill_stage[soff:soff + total].cpu().numpy()
                hseg.write(hblk.offset, cpu_bytes.tobytes())
                off = hblk.offset
                for p in parts:
                    nb = int(seg_bytes[p])
                    table.put(p, off, nb, hkey)
                    off += nb
                    self.metrics.bytes_written += nb
                soff += total
                blocks.append(hblk)
        self.metrics.records_written += n
        mgr.keep_alive(self.handle, self.map_id, blocks)
        mgr.publish_map_output(self.handle, self.map_id, table_addr)

    def _group_pairs_2level(self, hs, recs, n, W, key_bytes, R,
                            func, shift, nparts, dev, stream):
        """> 4096 partitions: the single radix pass is capped at 2^12 LDS
        digits, so the PARTITION ID itself is radixed in two levels —
        extract (pid, aux) pairs, scatter by pid>>12 (coarse), then per
        coarse bucket by pid&4095 (fine). Returns (full R counts, pairs,
        grouped pairs, gather digit params: pid is pair.x so the gather
        recomputes it as plain bits)."""
        import torch
        nbits = (R - 1).bit_length()
        if nbits > 24:
            raise ValueError(f"GPU partitioner supports R <= 2^24, got {R}")
        mask_full = (1 << nbits) - 1
        G = (R + 4095) >> 12
        nbits_c = max((G - 1).bit_length(), 4)
        ndc = 1 << nbits_c
        pairs = torch.empty(2 * n, dtype=torch.int64, device=dev)
        hs.extract_pairs(recs.data_ptr(), n, W, key_bytes, pairs.data_ptr(),
                         stream, func, shift, mask_full, nparts)
        hist = torch.empty(hs.radix_hist_bytes(n, 12) // 4,
                           dtype=torch.int32, device=dev)
        scan_ws = torch.empty(hs.radix_scan_ws_bytes(n, 12) // 4,
                              dtype=torch.int32, device=dev)
        totals_c = torch.empty(ndc, dtype=torch.int32, device=dev)
        # coarse: group by pid >> 12
        hs.radix_hist(pairs.data_ptr(), n, 12, nbits_c, hist.data_ptr(),
                      stream, 0, 2, 0)
        hs.radix_scan(hist.data_ptr(), n, nbits_c, totals_c.data_ptr(),
                      scan_ws.data_ptr(), stream)
        counts_c = totals_c.cpu().numpy().astype(np.int64)[:G]   # sync
        starts_c = np.zeros(G, dtype=np.int64)
        np.cumsum(counts_c[:-1], out=starts_c[1:])
        pairs_c = torch.empty_like(pairs)
        bases_c = torch.from_numpy(
            np.pad(starts_c, (0, ndc - G))).to(dev)
        kd = pairs_c.data_ptr() + bases_c * 16
        vd = kd + 8
        hs.radix_scatter(pairs.data_ptr(), pairs.data_ptr() + 8, n, 12,
                         nbits_c, hist.data_ptr(), kd.data_ptr(),
                         vd.data_ptr(), stream, 0, 1, 2, 0)
        # fine: per coarse bucket, group its subrange by pid & 4095.
        # All fine hists run first (stream-ordered, per-bucket totals
        # persist), ONE sync, then all fine scatters.
        totals_f = torch.empty(G * 4096, dtype=torch.int32, device=dev)
        bhists = []
        for b in range(G):
            cnt = int(counts_c[b])
            if cnt == 0:
                totals_f[b * 4096:(b + 1) * 4096] = 0
                bhists.append(None)
                continue
            bh = torch.empty(hs.radix_hist_bytes(cnt, 12) // 4,
                             dtype=torch.int32, device=dev)
            bws = torch.empty(hs.radix_scan_ws_bytes(cnt, 12) // 4,
                              dtype=torch.int32, device=dev)
            sub = pairs_c.data_ptr() + starts_c[b] * 16
            hs.radix_hist(sub, cnt, 0, 12, bh.data_ptr(), stream, 0, 2, 0)
            hs.radix_scan(bh.data_ptr(), cnt, 12,
                          totals_f.data_ptr() + b * 4096 * 4,
                          bws.data_ptr(), stream)
            bhists.append((bh, bws, sub, cnt))
        counts_full = totals_f.cpu().numpy().astype(np.int64)[:R]   # sync
        # pids b*4096+j with j beyond R's tail never occur; length R slice
        starts_full = np.zeros(R, dtype=np.int64)
        np.cumsum(counts_full[:-1], out=starts_full[1:])
        pairs_f = pairs   # reuse: original extraction no longer needed
        base_addr = pairs_f.data_ptr() + torch.from_numpy(
            starts_full).to(dev) * 16
        for b in range(G):
            if bhists[b] is None:
                continue
            bh, bws, sub, cnt = bhists[b]
            lo = b * 4096
            hi = min(lo + 4096, R)
            kd_b = torch.zeros(4096, dtype=torch.int64, device=dev)
            kd_b[:hi - lo] = base_addr[lo:hi]
            vd_b = kd_b + 8
            hs.radix_scatter(sub, sub + 8, cnt, 0, 12, bh.data_ptr(),
                             kd_b.data_ptr(), vd_b.data_ptr(), stream,
                             0, 1, 2, 0)
        # gather digit = pair.x (the pid) verbatim: func 0, shift 0
        return counts_full, pairs_c, pairs_f, 0, 0, mask_full, 0

    def _commit_gpu_records(self, partitioner) -> None:
        """Wide-record map-side GPU write. Records stay put while 16-byte
        (key-prefix, aux) pairs run the radix machinery; ONE gather pass
        then moves each W-byte record straight to its final HBM position
        (ops/csrc/kernels.hip 'wide-record machinery'). Replaces the
        reference's CPU writer + mmap + register for arbitrary-width
        records (RdmaMappedFile.java:113-189)."""
        import torch
        from .ops import load as ops_load
        mgr = self.manager
        if mgr.gpu is None:
            raise RuntimeError("GPU writer requires the GPU data plane")
        hs = ops_load()
        R = self.handle.num_partitions
        func, shift, nparts, nbits_eff = self._gpu_part_params(partitioner)
        W, key_bytes = self._record_shape
        batches = self._gpu_records
        recs = (batches[0] if len(batches) == 1 else torch.cat(batches))
        n = recs.numel() // W
        if n == 0:
            self._commit_empty()
            return
        dev = recs.device
        stream = torch.cuda.current_stream().cuda_stream
        two_level = (R - 1).bit_length() > 12
        if two_level:
            counts_nd, pairs, pairs_out, gfunc, gshift, gmask, gnparts = \
                self._group_pairs_2level(hs, recs, n, W, key_bytes, R,
                                         func, shift, nparts, dev, stream)
            nd = R
        else:
            nd = 1 << nbits_eff
            gfunc, gshift, gmask, gnparts = func, shift, nd - 1, nparts
            pairs = torch.empty(2 * n, dtype=torch.int64, device=dev)
            hs.extract_pairs(recs.data_ptr(), n, W, key_bytes,
                             pairs.data_ptr(), stream)
            hist = torch.empty(hs.radix_hist_bytes(n, nbits_eff) // 4,
                               dtype=torch.int32, device=dev)
            scan_ws = torch.empty(hs.radix_scan_ws_bytes(n, nbits_eff) // 4,
                                  dtype=torch.int32, device=dev)
            totals = torch.empty(nd, dtype=torch.int32, device=dev)
            hs.radix_hist(pairs.data_ptr(), n, shift, nbits_eff,
                          hist.data_ptr(), stream, func, 2, nparts)
            hs.radix_scan(hist.data_ptr(), n, nbits_eff, totals.data_ptr(),
                          scan_ws.data_ptr(), stream)
            counts_nd = totals.cpu().numpy().astype(np.int64)  # syncs
            pairs_out = None   # grouped below, after the layout
        counts = counts_nd[:R]
        seg_bytes = counts * W

        # greedy chunking of partitions into HBM blocks (same policy as
        # the 16-byte path / RdmaMappedFile.java:113-157)
        table, table_addr = mgr.alloc_table(R, self.handle.shuffle_id)
        write_block = mgr.conf.shuffle_write_block_size
        pool = mgr.gpu.pool
        blocks = []
        rec_dst = np.zeros(nd, dtype=np.int64)    # byte base per digit
        group, group_bytes = [], 0
        flushes = []
        for p in range(R):
            if group and group_bytes + seg_bytes[p] > write_block:
                flushes.append(group)
                group, group_bytes = [], 0
            group.append(p)
            group_bytes += int(seg_bytes[p])
        if group:
            flushes.append(group)
        meta_key = make_key(mgr.executor_id, 1)
        spill_groups = []   # (parts, total) that did not fit the HBM pool
        for parts in flushes:
            total = int(sum(seg_bytes[p] for p in parts))
            if total == 0:
                for p in parts:
                    table.put(p, 0, 0, meta_key)
                continue
            try:
                blk = pool.get(total)
            except MemoryError:
                spill_groups.append((parts, total))
                continue
            base = mgr.gpu.local_base(blk.segment_id)
            key = make_key(mgr.executor_id, blk.segment_id)
            off = blk.offset
            for p in parts:
                nb = int(seg_bytes[p])
                table.put(p, off, nb, key)
                if nb:
                    rec_dst[p] = base + off
                off += nb
                self.metrics.bytes_written += nb
            blocks.append(blk)
        spill_stage = None
        if spill_groups:
            stage_total = sum(t for _, t in spill_groups)
            spill_stage = torch.empty(stage_total, dtype=torch.uint8,
                                      device=dev)
            soff = 0
            for parts, total in spill_groups:
                for p in parts:
                    nb = int(seg_bytes[p])
                    if nb:
                        rec_dst[p] = spill_stage.data_ptr() + soff
                    soff += nb

        # group the PAIRS by digit (contiguous per-digit runs)
        starts = np.zeros(nd, dtype=np.int64)
        np.cumsum(counts_nd[:-1], out=starts[1:])
        if pairs_out is None:   # single-pass grouping (two-level grouped
            pairs_out = torch.empty_like(pairs)    # during count discovery)
            bases_t = torch.from_numpy(starts).to(dev)
            kd = pairs_out.data_ptr() + bases_t * 16
            vd = kd + 8
            hs.radix_scatter(pairs.data_ptr(), pairs.data_ptr() + 8, n,
                             shift, nbits_eff, hist.data_ptr(),
                             kd.data_ptr(), vd.data_ptr(), stream, func, 1,
                             2, nparts)
        # one gather moves every record to its final slot
        dstart_t = torch.from_numpy(starts.astype(np.uint32)
                                    .view(np.int32)).to(dev)
        dst_addr_t = torch.from_numpy(rec_dst).to(dev)
        hs.gather_records(recs.data_ptr(), pairs_out.data_ptr(), n, W, 1, 0,
                          dst_addr_t.data_ptr(), dstart_t.data_ptr(), gshift,
                          gmask, gfunc, gnparts, stream)
        torch.cuda.synchronize()
        if spill_groups:
            soff = 0
            for parts, total in spill_groups:
                hblk = mgr.pool.get(total)     # host pool
                hseg = mgr.data_segment(hblk.segment_id)
                hkey = make_key(mgr.executor_id, hblk.segment_id)
                cpu_bytes = spill_stage[soff:soff + total].cpu().numpy()
                hseg.write(hblk.offset, cpu_bytes.tobytes())
                off = hblk.offset
                for p in parts:
                    nb = int(seg_bytes[p])
                    table.put(p, off, nb, hkey)
                    off += nb
                    self.metrics.bytes_written += nb
                soff += total
                blocks.append(hblk)
        self.metrics.records_written += n
        mgr.keep_alive(self.handle, self.map_id, blocks)
        mgr.publish_map_output(self.handle, self.map_id, table_addr)

    def _commit_empty(self) -> None:
        """Zero-record map task: publish an all-empty location table."""
        mgr = self.manager
        R = self.handle.num_partitions
        table, table_addr = mgr.alloc_table(R, self.handle.shuffle_id)
        meta_key = make_key(mgr.executor_id, 1)
        for p in range(R):
            table.put(p, 0, 0, meta_key)
        mgr.keep_alive(self.handle, self.map_id, [])
        mgr.publish_map_output(self.handle, self.map_id, table_addr)

    def _commit_segments(self, segments: List[bytes]) -> None:
        """Greedy chunking at partition boundaries + table fill + publish."""
        mgr = self.manager
        write_block = mgr.conf.shuffle_write_block_size
        table, table_addr = mgr.alloc_table(self.handle.num_partitions,
                                    self.handle.shuffle_id)
        blocks = []
        # greedy grouping of partitions into blocks
        group: List[int] = []
        group_bytes = 0
        flushes: List[List[int]] = []
        for p, seg in enumerate(segments):
            if group and group_bytes + len(seg) > write_block:
                flushes.append(group)
                group, group_bytes = [], 0
            group.append(p)
            group_bytes += len(seg)
        if group:
            flushes.append(group)
        meta_key = make_key(mgr.executor_id, 1)  # META_SEGMENT_ID
        for parts in flushes:
            total = sum(len(segments[p]) for p in parts)
            if total == 0:
                for p in parts:
                    table.put(p, 0, 0, meta_key)
                continue
            blk = mgr.pool.get(total)
            seg = mgr.data_segment(blk.segment_id)
            key = make_key(mgr.executor_id, blk.segment_id)
            off = blk.offset
            for p in parts:
                data = segments[p]
                if data:
                    seg.write(off, data)
                table.put(p, off, len(data), key)
                off += len(data)
                self.metrics.bytes_written += len(data)
            blocks.append(blk)
        mgr.keep_alive(self.handle, self.map_id, blocks)
        mgr.publish_map_output(self.handle, self.map_id, table_addr)
