"""SQL sort-merge join — BASELINE config 5 (1B x 1B rows on 8 GPUs).

Two synthetic tables of (key u64, payload u64) rows. One step:

  1. shuffle table A by key range into R global partitions (GPU radix
     partition straight into HBM blocks)
  2. shuffle table B likewise (second shuffle)
  3. per owned partition: radix-sort both sides over the low bits, then
     the hand-written merge kernel (join_count/join_emit) produces the
     matched pairs

Keys are drawn from a shared space so a controllable fraction matches;
validation checks the exact match count against a CPU oracle.
"""

from __future__ import annotations

import time
from dataclasses import dataclass

import numpy as np

from ..engine import Engine
from ..partitioner import RangePartitioner


@dataclass
class JoinResult:
    seconds: float
    rows_a: int
    rows_b: int
    matches: int
    shuffle_bytes: int


class SortMergeJoin:
    def __init__(self, engine: Engine, rows_per_executor: int,
                 partitions_per_executor: int = 64, device: str = "cpu",
                 key_space_bits: int = 32, validate: bool = False,
                 seed: int = 0):
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
        self.key_bits = key_space_bits
        self.part = RangePartitioner.uniform(
            R, key_min=0, key_max=(1 << key_space_bits) - 1)
        self.low_bits = key_space_bits - (R - 1).bit_length()
        rank = engine.rank
        rng = np.random.default_rng(seed * 31 + rank)
        ka = rng.integers(0, 1 << key_space_bits, self.n, dtype=np.uint64)
        kb = rng.integers(0, 1 << key_space_bits, self.n, dtype=np.uint64)
        if device == "cuda":
            import torch
            self.a_keys = torch.from_numpy(ka.view(np.int64)).cuda()
            self.b_keys = torch.from_numpy(kb.view(np.int64)).cuda()
            self.a_vals = self.a_keys.clone()   # payload := key (checkable)
            self.b_vals = self.b_keys.clone()
        else:
            self.a_keys, self.b_keys = ka, kb
            self.a_vals = ka.view(np.uint8).reshape(-1, 8).copy()
            self.b_vals = kb.view(np.uint8).reshape(-1, 8).copy()

    def _shuffle_sorted(self, keys, vals):
        """One table's shuffle, consumed through the reader's GENERIC
        ordering hookup (read_aos(ordering=True) — the shared
        deserialize->sort path, RdmaShuffleReader.scala:61-114 role):
        returns (handle, sorted keys, sorted vals, metrics)."""
        eng = self.engine
        handle = eng.register_shuffle(eng.world_size, self.R)
        w = eng.manager.get_writer(handle, eng.rank)
        if self.device == "cuda":
            w.write_device_batch(keys, vals)
        else:
            w.write_batch(keys, vals)
        w.stop(True, partitioner=self.part)
        eng.barrier()
        lo, hi = eng.rank * self.ppe, (eng.rank + 1) * self.ppe - 1
        arena_hint = None
        if self.device == "cuda":
            import torch
            cap = int(self.n * 16 * 1.25) + (64 << 10)
            self._arena_pool = getattr(self, "_arena_pool", [])
            for t in self._arena_pool:
                if not getattr(t, "_in_use", False) and t.numel() >= cap:
                    arena_hint = t
                    break
            if arena_hint is None:
                arena_hint = torch.empty(cap, dtype=torch.uint8,
                                         device="cuda")
                self._arena_pool.append(arena_hint)
            # table A's data must survive table B's shuffle
            arena_hint._in_use = True
        reader = eng.manager.get_reader(handle, lo, hi, arena=arena_hint)
        # the owned partition range shares its top log2(W) key bits, so
        # ONE batched sort over the low bits fully orders each side
        wbits = (eng.world_size - 1).bit_length()
        k, v = reader.read_aos(ordering=True,
                               end_bit=self.key_bits - wbits)
        return handle, k, v, reader.metrics

    def run_step(self) -> JoinResult:
        eng = self.engine
        t0 = time.perf_counter()
        ha, ak, av, ma = self._shuffle_sorted(self.a_keys, self.a_vals)
        hb, bk, bv, mb = self._shuffle_sorted(self.b_keys, self.b_vals)
        matches = 0
        if self.device == "cuda":
            import torch
            from ..ops.join import merge_join_sorted
            if ak.numel() and bk.numel():
                jk, ja, jb = merge_join_sorted(
                    ak.contiguous(), av.contiguous(),
                    bk.contiguous(), bv.contiguous())
                matches += jk.numel()
                if self.validate and jk.numel():
                    assert torch.equal(jk, ja) and torch.equal(jk, jb), \
                        "joined payloads must equal keys"
            torch.cuda.synchronize()
        elif len(ak) and len(bk):
            lo = np.searchsorted(bk, ak, side="left")
            hi = np.searchsorted(bk, ak, side="right")
            matches += int((hi - lo).sum())
        eng.unregister_shuffle(ha)
        eng.unregister_shuffle(hb)
        for t in getattr(self, "_arena_pool", []):
            t._in_use = False
        dt = time.perf_counter() - t0
        if self.validate:
            self._validate_counts(matches)
        return JoinResult(dt, self.n, self.n, matches,
                          ma.remote_bytes_read + mb.remote_bytes_read)

    def _validate_counts(self, matches: int) -> None:
        """Local-count oracle only works single-process; multi-rank runs
        validate via all-rank sum equality in the caller."""
        if self.engine.world_size != 1:
            return
        if self.device == "cuda":
            ka = self.a_keys.cpu().numpy().view(np.uint64)
            kb = self.b_keys.cpu().numpy().view(np.uint64)
        else:
            ka, kb = self.a_keys, self.b_keys
        bs = np.sort(kb)
        want = int((np.searchsorted(bs, ka, "right")
                    - np.searchsorted(bs, ka, "left")).sum())
        assert matches == want, f"join matches {matches} != oracle {want}"
