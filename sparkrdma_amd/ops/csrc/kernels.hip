// CDNA4 (gfx950) shuffle kernels: radix bucket-scatter machinery.
//
// One primitive serves both hot paths (SURVEY.md §2.3 "new GPU kernels"):
//   * map-side partition+serialize: digit = top log2(R) key bits; the
//     scatter writes keys/values STRAIGHT into their final positions in
//     HBM block buffers (per-digit destination base pointers) — no
//     intermediate file, replacing Spark's CPU sort-shuffle writer
//     (reference RdmaWrapperShuffleWriter.scala:83-102).
//   * reduce-side LSD radix sort: 8-bit digits, ping-pong buffers.
//
// Design (MI355X-first, not a CUDA port):
//   * wave = 64: rank computation is a 64-lane ballot multi-split
//     (8..12 __ballot()s of 64 bits), per-wave LDS counters — no atomics.
//   * stability: wave w owns the contiguous element chunk
//     [tile + w*64*ITEMS, ...), iterated lane-major, so block order ==
//     memory order and ranks are stable.
//   * scattered global writes are coalesced through an LDS exchange
//     (elements digit-sorted in LDS, then written out linearly — each
//     digit's run is a contiguous global write burst).
//   * partition path: 3-kernel (hist -> hierarchical scan -> scatter;
//     row-major [nb][ND] hist so every access is thread==digit coalesced)
//     because the host must see counts to lay out HBM blocks first.
//   * sort path: onesweep — one kernel per pass with decoupled lookback
//     (u64 flag|count agent-scope descriptors), AoS records, deferred
//     walk; selectable hist+scan mode kept for A/B.
//
// Limits: n < 2^32 elements per call; ONE pass covers 2^4..2^12 LDS
// digits — arbitrary partition counts come from the partition funcs
// (PartFunc 2/3) and, above 4096, the writer's two-level pid radix.

#include "common.h"

#include <hip/hip_runtime.h>

namespace hipshuffle {

constexpr int BLOCK = 256;        // 4 waves
constexpr int ITEMS = 16;         // elements per thread
constexpr int TILE = BLOCK * ITEMS;  // 4096 elements per workgroup
constexpr int NW = BLOCK / kWave;    // waves per block

__device__ __forceinline__ uint32_t wave_inclusive_scan(uint32_t v) {
#pragma unroll
  for (int d = 1; d < kWave; d <<= 1) {
    uint32_t t = __shfl_up(v, d);
    if ((threadIdx.x & (kWave - 1)) >= (unsigned)d) v += t;
  }
  return v;
}

// Exclusive scan of arr[nd] in LDS (nd pow2); sums: >=BS u32.
template <int BS = BLOCK>
__device__ void block_exscan(uint32_t* arr, uint32_t* sums, int nd) {
  const int tid = threadIdx.x;
  const int per = (nd + BS - 1) / BS;  // >= 1
  const int lo = tid * per;
  uint32_t run = 0;
#pragma unroll 1
  for (int k = 0; k < per; ++k) {
    int idx = lo + k;
    uint32_t v = idx < nd ? arr[idx] : 0;
    if (idx < nd) arr[idx] = run;
    run += v;
  }
  sums[tid] = run;
  __syncthreads();
  if (tid < kWave) {
    uint32_t off = 0;
#pragma unroll 1
    for (int c = 0; c < BS / kWave; ++c) {
      uint32_t v = sums[c * kWave + tid];
      uint32_t inc = wave_inclusive_scan(v);
      sums[c * kWave + tid] = off + inc - v;
      off += __shfl(inc, kWave - 1);
    }
  }
  __syncthreads();
  uint32_t base = sums[tid];
#pragma unroll 1
  for (int k = 0; k < per; ++k) {
    int idx = lo + k;
    if (idx < nd) arr[idx] += base;
  }
  __syncthreads();
}

// splitmix64 finalizer step — bit-identical to partitioner.HashPartitioner
// so GPU hash partitioning matches the CPU oracle exactly.
__device__ __forceinline__ uint64_t hash_mix64(uint64_t k) {
  k ^= k >> 33;
  k *= 0xFF51AFD7ED558CCDull;
  k ^= k >> 33;
  return k;
}

// Partition functions (the `func` runtime param of hist/scatter). 0/1 are
// the r01 bit-extraction modes; 2/3 lift the pow2-only restriction
// (VERDICT r01 missing item 7): any R <= 2^12 via multiply-high range
// partitioning (pid = floor(key * R / 2^64), exactly the CPU
// RangePartitioner with ceil bounds) or hash-mod. The digit still fits
// the pow2 LDS counter layout (ND = next pow2 >= R).
enum PartFunc : int {
  kPartBits = 0,      // (key >> shift) & mask
  kPartHashBits = 1,  // (mix(key) >> shift) & mask   (== mod for pow2 R)
  kPartRange = 2,     // mulhi(key, R)                 (any R)
  kPartHashMod = 3,   // mix(key) % R                  (any R)
};

__device__ __forceinline__ uint32_t digit_of(uint64_t k, int shift,
                                             uint32_t mask, int func,
                                             uint32_t nparts) {
  switch (func) {
    default:
    case kPartBits: return (uint32_t)(k >> shift) & mask;
    case kPartHashBits: return (uint32_t)(hash_mix64(k) >> shift) & mask;
    case kPartRange:
      return (uint32_t)(((unsigned __int128)k * nparts) >> 64);
    case kPartHashMod: return (uint32_t)(hash_mix64(k) % nparts);
  }
}

// Match from PRE-COMPUTED per-bit ballots (register votes) — lets rank
// logic reuse one iteration's ballots against another iteration's digits.
template <int PBITS>
__device__ __forceinline__ uint64_t match_votes(const uint64_t* v,
                                                uint64_t vm, uint32_t d) {
  uint64_t m = vm;
#pragma unroll
  for (int b = 0; b < PBITS; ++b) m &= ((d >> b) & 1) ? v[b] : ~v[b];
  return m;
}

// 64-lane multi-split: lanes with equal `digit` (among `validmask` lanes).
template <int NBITS>
__device__ __forceinline__ uint64_t match_lanes(uint32_t digit,
                                                uint64_t validmask) {
  uint64_t match = validmask;
#pragma unroll
  for (int b = 0; b < NBITS; ++b) {
    uint64_t vote = __ballot((digit >> b) & 1);
    match &= ((digit >> b) & 1) ? vote : ~vote;
  }
  return match;
}

// ---------------------------------------------------------------------------
// Kernel 1: per-block digit histogram. hist layout: [nb][ND] (row-major) —
// hist writes, scan passes and the scatter's prefix preload are all
// coalesced across the 256 threads (thread==digit lane).

template <int NBITS>
__global__ __launch_bounds__(BLOCK) void radix_hist_kernel(
    const uint64_t* __restrict__ keys, uint32_t n, int shift,
    uint32_t* __restrict__ hist, uint32_t nb, int func,
    int in_stride = 1, uint32_t nparts = 0) {
  constexpr int ND = 1 << NBITS;
  extern __shared__ char smem_raw[];
  uint32_t* counters = reinterpret_cast<uint32_t*>(smem_raw);  // [NW][ND]
  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wave = tid >> 6;
  const uint32_t b = blockIdx.x;
  const uint64_t tile_start = (uint64_t)b * TILE;
  uint32_t* my = counters + wave * ND;
  for (int d = tid; d < NW * ND; d += BLOCK) counters[d] = 0;
  __syncthreads();

  const uint64_t chunk = tile_start + (uint64_t)wave * (ITEMS * kWave);
  // loads issued in their own unrolled loop so they are all in flight
  // before the serial LDS counter chain (same fix as the sort pass:
  // phase A 11.4 -> 8.2 us/block)
  uint64_t key_reg[ITEMS];
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    uint64_t e = chunk + (uint64_t)i * kWave + lane;
    key_reg[i] = e < n ? keys[e * in_stride] : 0;
  }
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    uint64_t e = chunk + (uint64_t)i * kWave + lane;
    bool valid = e < n;
    uint32_t d = digit_of(key_reg[i], shift, ND - 1, func, nparts);
    uint64_t vm = __ballot(valid);
    if (valid) {
      uint64_t match = match_lanes<NBITS>(d, vm);
      uint64_t lt = (1ull << lane) - 1;
      if ((match & lt) == 0) my[d] += (uint32_t)__popcll(match);
    }
  }
  __syncthreads();
  for (int d = tid; d < ND; d += BLOCK) {
    uint32_t s = 0;
#pragma unroll
    for (int w = 0; w < NW; ++w) s += counters[w * ND + d];
    hist[(uint64_t)b * ND + d] = s;
  }
}

// ---------------------------------------------------------------------------
// Kernel 2: hierarchical scan of hist[nb][ND] along the block axis per
// digit. Three coalesced passes (thread == digit lane, rows iterated
// serially): K2a per-chunk digit sums -> K2b single-block scan over
// chunks (+ digit totals) -> K2c rewrite rows as running prefixes.
// Cross-chunk sums travel through global memory between kernels — kernel
// boundaries are the release/acquire, no in-launch cross-XCD protocol
// needed. The matrix is ~2% of a pass's traffic; the naive per-column
// chained scan was O(nb) *latency* (4.5 ms at nb=655k).

template <int ND>
__global__ __launch_bounds__(BLOCK) void scan_chunk_sums_kernel(
    const uint32_t* __restrict__ hist, uint32_t nb, uint32_t chunk_rows,
    uint32_t* __restrict__ partial) {
  const uint32_t c = blockIdx.x;
  const uint32_t lo = c * chunk_rows, hi = min(lo + chunk_rows, nb);
  for (int d = threadIdx.x; d < ND; d += BLOCK) {
    uint32_t run = 0;
    for (uint32_t r = lo; r < hi; ++r) run += hist[(uint64_t)r * ND + d];
    partial[(uint64_t)c * ND + d] = run;
  }
}

template <int ND>
__global__ __launch_bounds__(BLOCK) void scan_chunks_kernel(
    uint32_t* __restrict__ partial, uint32_t chunks,
    uint32_t* __restrict__ totals) {
  for (int d = threadIdx.x; d < ND; d += BLOCK) {
    uint32_t run = 0;
    for (uint32_t c = 0; c < chunks; ++c) {
      uint32_t v = partial[(uint64_t)c * ND + d];
      partial[(uint64_t)c * ND + d] = run;
      run += v;
    }
    totals[d] = run;
  }
}

template <int ND>
__global__ __launch_bounds__(BLOCK) void scan_rewrite_kernel(
    uint32_t* __restrict__ hist, uint32_t nb, uint32_t chunk_rows,
    const uint32_t* __restrict__ partial) {
  const uint32_t c = blockIdx.x;
  const uint32_t lo = c * chunk_rows, hi = min(lo + chunk_rows, nb);
  for (int d = threadIdx.x; d < ND; d += BLOCK) {
    uint32_t run = partial[(uint64_t)c * ND + d];
    for (uint32_t r = lo; r < hi; ++r) {
      uint32_t v = hist[(uint64_t)r * ND + d];
      hist[(uint64_t)r * ND + d] = run;
      run += v;
    }
  }
}

static inline uint32_t scan_chunk_rows(uint32_t nb) {
  // target ~2048 blocks; at least 64 rows per chunk to amortize
  uint32_t c = (nb + 2047) / 2048;
  return c < 64 ? 64 : c;
}

// ---------------------------------------------------------------------------
// Kernel 2b (sort path): turn totals into per-digit destination pointers
// inside the ping-pong output buffers — keeps the whole sort device-side
// (no host round trip between passes).

template <int NBITS>
__global__ void radix_digit_bases_kernel(const uint32_t* __restrict__ totals,
                                         uint64_t out_keys, uint64_t out_vals,
                                         uint64_t* __restrict__ key_dst,
                                         uint64_t* __restrict__ val_dst) {
  constexpr int ND = 1 << NBITS;
  __shared__ uint32_t arr[ND];
  __shared__ uint32_t sums[BLOCK];
  const int tid = threadIdx.x;
  for (int d = tid; d < ND; d += BLOCK) arr[d] = totals[d];
  __syncthreads();
  block_exscan(arr, sums, ND);
  for (int d = tid; d < ND; d += BLOCK) {
    key_dst[d] = out_keys + (uint64_t)arr[d] * 8;
    val_dst[d] = out_vals ? out_vals + (uint64_t)arr[d] * 8 : 0;
  }
}

// ---------------------------------------------------------------------------
// Kernel 3: rank + LDS exchange + scatter to per-digit destinations.

template <int NBITS, bool HAS_VAL>
__global__ __launch_bounds__(BLOCK) void radix_scatter_kernel(
    const uint64_t* __restrict__ keys, const uint64_t* __restrict__ vals,
    uint32_t n, int shift, const uint32_t* __restrict__ hist, uint32_t nb,
    const uint64_t* __restrict__ key_dst, const uint64_t* __restrict__ val_dst,
    int func, int aos_out, int in_stride = 1, uint32_t nparts = 0) {
  constexpr int ND = 1 << NBITS;
  extern __shared__ char smem_raw[];
  // layout: exchange u64[TILE] | counters u32[NW][ND] | start u32[ND]
  //         | pref u32[ND] | sums u32[BLOCK]
  uint64_t* exch = reinterpret_cast<uint64_t*>(smem_raw);
  uint32_t* counters = reinterpret_cast<uint32_t*>(exch + TILE);
  uint32_t* start = counters + NW * ND;
  uint32_t* pref = start + ND;
  uint32_t* sums = pref + ND;

  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wave = tid >> 6;
  const uint32_t b = blockIdx.x;
  const uint64_t tile_start = (uint64_t)b * TILE;
  const uint32_t tile_n =
      (uint32_t)min((uint64_t)TILE, (uint64_t)n - tile_start);
  uint32_t* my = counters + wave * ND;

  for (int d = tid; d < NW * ND; d += BLOCK) counters[d] = 0;
  // preload this block's cross-block digit prefixes (coalesced row read)
  for (int d = tid; d < ND; d += BLOCK) pref[d] = hist[(uint64_t)b * ND + d];
  __syncthreads();

  // phase A: per-element (digit, rank within wave&digit), stable;
  // loads split from the serial LDS counter chain (see sort pass)
  uint64_t key_reg[ITEMS];
  uint32_t rank_reg[ITEMS];
  uint32_t dig_reg[ITEMS];
  const uint64_t chunk = tile_start + (uint64_t)wave * (ITEMS * kWave);
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    uint64_t e = chunk + (uint64_t)i * kWave + lane;
    key_reg[i] = e < n ? keys[e * in_stride] : 0;
  }
#pragma unroll
  for (int i = 0; i < ITEMS; ++i) {
    uint64_t e = chunk + (uint64_t)i * kWave + lane;
    bool valid = e < n;
    uint32_t d = digit_of(key_reg[i], shift, ND - 1, func, nparts);
    dig_reg[i] = d;
    uint64_t vm = __ballot(valid);
    uint32_t r = 0;
    if (valid) {
      uint64_t match = match_lanes<NBITS>(d, vm);
      uint64_t lt = (1ull << lane) - 1;
      uint32_t rank_in_iter = (uint32_t)__popcll(match & lt);
      uint32_t c = my[d];
      r = c + rank_in_iter;
      if (rank_in_iter == 0) my[d] = c + (uint32_t)__popcll(match);
    }
    rank_reg[i] = r;
  }
  __syncthreads();

  // scan waves per digit; block totals into start[]
  for (int d = tid; d < ND; d += BLOCK) {
    uint32_t run = 0;
#pragma unroll
    for (int w = 0; w < NW; ++w) {
      uint32_t t = counters[w * ND + d];
      counters[w * ND + d] = run;
      run += t;
    }
    start[d] = run;
  }
  __syncthreads();
  block_exscan(start, sums, ND);  // start[d] = block-local digit start

  // LDS exchange: keys to digit-sorted local order
#pragma unroll 1
  for (int i = 0; i < ITEMS; ++i) {
    uint64_t e = chunk + (uint64_t)i * kWave + lane;
    if (e < n) {
      uint32_t d = dig_reg[i];
      uint32_t j = start[d] + my[d] + rank_reg[i];
      rank_reg[i] = j;  // reuse: now holds local sorted position
      exch[j] = key_reg[i];
    }
  }
  __syncthreads();

  // linear write-out of keys; remember (digit, global offset) per slot
  uint32_t out_d[ITEMS];
  uint32_t out_off[ITEMS];
#pragma unroll 1
  for (int i = 0; i < ITEMS; ++i) {
    uint32_t j = i * BLOCK + tid;
    if (j < tile_n) {
      uint64_t k = exch[j];
      uint32_t d = digit_of(k, shift, ND - 1, func, nparts);
      uint32_t off = (pref[d] + (j - start[d])) << aos_out;
      out_d[i] = d;
      out_off[i] = off;
      reinterpret_cast<uint64_t*>(key_dst[d])[off] = k;
    }
  }
  if (HAS_VAL) {
    __syncthreads();
#pragma unroll 1
    for (int i = 0; i < ITEMS; ++i) {
      uint64_t e = chunk + (uint64_t)i * kWave + lane;
      if (e < n) exch[rank_reg[i]] = vals[e * in_stride];
    }
    __syncthreads();
#pragma unroll 1
    for (int i = 0; i < ITEMS; ++i) {
      uint32_t j = i * BLOCK + tid;
      if (j < tile_n)
        reinterpret_cast<uint64_t*>(val_dst[out_d[i]])[out_off[i]] = exch[j];
    }
  }
}

// ---------------------------------------------------------------------------
// Onesweep path (8-bit digits): one kernel per pass with decoupled
// lookback — removes the per-pass hist read and the scan kernels.
//
// Status word per (block, digit): 2-bit flag | 62-bit count/prefix packed
// in ONE u64, so a single relaxed agent-scope atomic is a complete
// publish (no multi-location ordering; the guide's G16 concerns collapse
// to single-word atomicity; an 8-byte relaxed agent store lowers to an
// sc1 write-through). Agent-scope atomics are cache-coherent across
// XCDs. Blocks take an atomic ticket at start, so a block's predecessors
// in lookback order are already launched -> no deadlock, and the ticket
// determines the tile, so results are placement- and timing-independent.

constexpr uint64_t FLAG_AGG = 1ull << 62;
constexpr uint64_t FLAG_INC = 2ull << 62;
constexpr uint64_t FLAG_MASK = 3ull << 62;
constexpr uint64_t VAL_MASK = (1ull << 62) - 1;

// Serial decoupled-lookback walk for digit `tid`: sum predecessors'
// aggregates back to the nearest inclusive prefix, publish our own
// inclusive, record the exclusive prefix in pref[tid]. (A windowed
// 8-wide variant measured NO faster — the cost is waiting on the
// publish frontier, not load throughput — so the caller overlaps this
// wait with the LDS exchange instead.)
template <int ND>
__device__ __forceinline__ void lookback_walk(
    uint64_t* __restrict__ desc, uint32_t b, int tid, uint64_t my_total,
    uint32_t* __restrict__ pref) {
  uint64_t run = 0;
  for (int64_t j = (int64_t)b - 1; j >= 0;) {
    uint64_t v = __hip_atomic_load(&desc[(uint64_t)j * ND + tid],
                                   __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
    if ((v & FLAG_MASK) == 0) {
      __builtin_amdgcn_s_sleep(2);
      continue;
    }
    run += v & VAL_MASK;
    if ((v & FLAG_MASK) == FLAG_INC) break;
    --j;
  }
  __hip_atomic_store(&desc[(uint64_t)b * ND + tid],
                     FLAG_INC | (run + my_total), __ATOMIC_RELAXED,
                     __HIP_MEMORY_SCOPE_AGENT);
  pref[tid] = (uint32_t)run;
}

// TRANSPOSED-descriptor walk: desc laid out [ND][nb] so digit `tid`'s
// predecessors are CONTIGUOUS backwards — 16 descriptors share one
// 128-B line, so a completed prefix region costs nb/16 line fills
// instead of nb (the r01 [nb][ND] layout made every probe its own
// line). Publish is a scattered 8-B store per digit (one line each) —
// worth it when walks span many predecessors (512-block resident
// windows at 64M records).
template <int ND>
__device__ __forceinline__ void lookback_walk_t(
    uint64_t* __restrict__ desc, uint32_t nb, uint32_t b, int tid,
    uint64_t my_total, uint32_t* __restrict__ pref) {
  uint64_t* row = desc + (uint64_t)tid * nb;
  uint64_t run = 0;
  for (int64_t j = (int64_t)b - 1; j >= 0;) {
    uint64_t v = __hip_atomic_load(&row[j], __ATOMIC_RELAXED,
                                   __HIP_MEMORY_SCOPE_AGENT);
    if ((v & FLAG_MASK) == 0) {
      __builtin_amdgcn_s_sleep(2);
      continue;
    }
    run += v & VAL_MASK;
    if ((v & FLAG_MASK) == FLAG_INC) break;
    --j;
  }
  __hip_atomic_store(&row[b], FLAG_INC | (run + my_total),
                     __ATOMIC_RELAXED, __HIP_MEMORY_SCOPE_AGENT);
  pref[tid] = (uint32_t)run;
}

// Global digit totals of EVERY pass in one read (digit counts are
// order-independent, so pass k's totals can be computed from pass 0's
// input).
template <int MAX_PASSES, int PBITS>
__global__ __launch_bounds__(BLOCK) void onesweep_hist_all_kernel(
    const uint64_t* __restrict__ keys, uint32_t n, int start_bit, int passes,
    uint32_t* __restrict__ totals /* [passes][1<<PBITS] */, int in_stride,
    int word_off = 0) {
  constexpr int PD = 1 << PBITS;
  __shared__ uint32_t cnt[NW][MAX_PASSES * PD];
  const int tid = threadIdx.x;
  const int wave = tid >> 6;
  for (int i = tid; i < NW * MAX_PASSES * PD; i += BLOCK)
    cnt[0][i] = 0;  // flat zero (cnt rows contiguous)
  __syncthreads();
  const uint64_t stride = (uint64_t)gridDim.x * BLOCK;
  for (uint64_t e = (uint64_t)blockIdx.x * BLOCK + tid; e < n; e += stride) {
    uint64_t k = keys[e * in_stride + word_off];
    for (int p = 0; p < passes; ++p) {
      int sb = start_bit + p * PBITS;
      if (sb > 64 - PBITS) sb = 64 - PBITS;  // last pass re-covers top bits
      uint32_t d = (uint32_t)(k >> sb) & (PD - 1);
      atomicAdd(&cnt[wave][p * PD + d], 1);
    }
  }
  __syncthreads();
  for (int i = tid; i < passes * PD; i += BLOCK) {
    uint32_t s = 0;
#pragma unroll
    for (int w = 0; w < NW; ++w) s += cnt[w][i];
    if (s) atomicAdd(&totals[i], s);
  }
}

// AOS=true: elements are interleaved (key, val) 16-byte records — one
// dwordx4 load per element, a pair LDS exchange, ONE 16-byte store per
// element, and 2x-longer digit write bursts (measured: scattered-write
// bandwidth doubles from 128 B to 256 B bursts — profiles/).
// LEAN (AoS only): phase A loads ONLY the sort word and the exchange
// RE-READS the full pair from global (L2-hot: the tile was just pulled)
// — drops the val_reg/key_reg liveness so VGPRs fit 6 waves/SIMD =
// 3 blocks/CU resident, hiding more of the lookback wait while keeping
// full-tile burst lengths (the 2048-tile route paid burst halving +
// 2x descriptors for its occupancy; this pays only L1/L2 re-reads).
template <bool HAS_VAL, int IT, bool AOS, int BS = BLOCK, int PBITS = 8,
          bool LEAN = false, bool FUSE_GATHER = false>
__global__ __launch_bounds__(BS)
__attribute__((amdgpu_waves_per_eu(LEAN ? 6 : 1))) void onesweep_pass_kernel(
    const uint64_t* __restrict__ keys, const uint64_t* __restrict__ vals,
    uint32_t n, int shift, uint64_t* __restrict__ desc /* [nb][ND] */,
    uint32_t* __restrict__ ticket,
    const uint64_t* __restrict__ key_dst, const uint64_t* __restrict__ val_dst,
    int stage = 3, uint64_t* __restrict__ timing = nullptr,
    const uint32_t* __restrict__ hist_pref = nullptr, int sort_word = 0,
    int lb_mode = 0 /* 0: [nb][ND] descriptors, 1: transposed [ND][nb] */,
    int split_exch = 0, /* AoS: exchange+writeout in N sub-tile rounds
                           (0/1 = single round); the LDS exchange buffer
                           shrinks by N so more blocks stay resident to
                           hide the lookback wait */
    const uint32_t* __restrict__ rec_words = nullptr, /* FUSE_GATHER */
    uint32_t rec_w4 = 0 /* record words; key_dst then holds per-digit
                           RECORD base addresses and the writeout copies
                           whole records instead of pairs — the final
                           sort pass IS the gather */) {
  constexpr int ND = 1 << PBITS;
  extern __shared__ char smem_raw[];
  constexpr int TILE_T = BS * IT;
  constexpr int NWT = BS / kWave;
  using u64x2 = __attribute__((ext_vector_type(2))) unsigned long long;
  const int ecap = (AOS && split_exch > 1) ? TILE_T / split_exch : TILE_T;
  uint64_t* exch = reinterpret_cast<uint64_t*>(smem_raw);  // [ecap] or
  u64x2* exch2 = reinterpret_cast<u64x2*>(smem_raw);       // [ecap] pairs
  uint32_t* counters = reinterpret_cast<uint32_t*>(
      exch + (AOS ? 2 * ecap : TILE_T));
  uint32_t* start = counters + NWT * ND;
  uint32_t* pref = start + ND;
  uint32_t* sums = pref + ND;  // [BS]
  uint32_t* vb_sh = sums + BS;
  uint8_t* dsort = reinterpret_cast<uint8_t*>(vb_sh + 4);  // [TILE_T], !AOS

  const int tid = threadIdx.x;
  const int lane = tid & (kWave - 1);
  const int wave = tid >> 6;
  if (tid == 0)
    *vb_sh = hist_pref ? blockIdx.x : atomicAdd(ticket, 1);
  for (int d = tid; d < NWT * ND; d += BS) counters[d] = 0;
  __syncthreads();
  const uint32_t b = *vb_sh;  // execution-ordered virtual block id
  const uint64_t tile_start = (uint64_t)b * TILE_T;
  const uint32_t tile_n =
      (uint32_t)min((uint64_t)TILE_T, (uint64_t)n - tile_start);
  uint32_t* my = counters + wave * ND;

  uint64_t t0 = timing ? __builtin_amdgcn_s_memrealtime() : 0;
  // phase A: stable per-wave ranks; digit(8b)|rank(16b) packed per elem.
  // Loads run in their own loop FIRST so all IT 16-byte loads are in
  // flight before the serial per-wave LDS counter chain starts — fusing
  // load+rank serialized loads behind LDS RMWs (phase A 11.4 -> 8.2 us
  // per block, full sort 7.56 -> 7.15 ms / 64M).
  uint64_t key_reg[(AOS && LEAN) ? 1 : IT];
  uint64_t val_reg[(AOS && !LEAN) ? IT : 1];
  uint32_t digrank[IT];
  const uint64_t chunk = tile_start + (uint64_t)wave * (IT * kWave);
  if (!(AOS && LEAN)) {
#pragma unroll
    for (int i = 0; i < IT; ++i) {
      uint64_t e = chunk + (uint64_t)i * kWave + lane;
      bool valid = e < n;
      if (AOS) {
        u64x2 kv = valid ? reinterpret_cast<const u64x2*>(keys)[e]
                         : u64x2{0, 0};
        key_reg[i] = kv.x;
        val_reg[(AOS && !LEAN) ? i : 0] = kv.y;
      } else {
        key_reg[i] = valid ? keys[e] : 0;
      }
    }
  }
  // rank in iteration PAIRS: both iterations' per-bit ballots are held
  // in registers, so the pair's ranks and the single LDS counter update
  // per digit need only ONE serial LDS round trip per two iterations —
  // the serial chain was 13% of a pass (stage-9 ablation).
  static_assert(IT % 2 == 0, "pair ranking needs even IT");
  const uint64_t lt = (1ull << lane) - 1;
  if (AOS && LEAN) {
    // single-iteration ranking: one ballot set live at a time — the
    // VGPR budget (not LDS) is what buys the 3rd resident block per CU;
    // the serial LDS chain it re-lengthens is hidden by that occupancy
    const int sw = sort_word ? 1 : 0;
#pragma unroll
    for (int i = 0; i < IT; ++i) {
      uint64_t e = chunk + (uint64_t)i * kWave + lane;
      bool valid = e < n;
      uint64_t k = valid ? keys[2 * e + sw] : 0;
      uint32_t d = (uint32_t)(k >> shift) & (ND - 1);
      uint64_t vm = __ballot(valid);
      uint32_t r = 0;
      if (valid) {
        uint64_t match = match_lanes<PBITS>(d, vm);
        uint32_t rank_in_iter = (uint32_t)__popcll(match & lt);
        uint32_t base = my[d];
        r = base + rank_in_iter;
        if (rank_in_iter == 0) my[d] = base + (uint32_t)__popcll(match);
      }
      digrank[i] = (d << 16) | (valid ? r : 0);
    }
  } else
#pragma unroll
  for (int c = 0; c < IT / 2; ++c) {
    const int i0 = 2 * c, i1 = 2 * c + 1;
    uint64_t e0 = chunk + (uint64_t)i0 * kWave + lane;
    uint64_t e1 = chunk + (uint64_t)i1 * kWave + lane;
    bool val0 = e0 < n, val1 = e1 < n;
    // sort_word selects WHICH u64 of the AoS pair orders this pass
    // (the 80-bit-key path sorts 16 aux bits before the 64 prefix bits).
    // LEAN keeps NO register arrays: the sort word loads here and the
    // exchange re-reads the pair from the L2-hot tile — the VGPR budget
    // is what buys the 3rd resident block per CU.
    uint64_t sk0, sk1;
    if (AOS && LEAN) {
      const int sw = sort_word ? 1 : 0;
      sk0 = val0 ? keys[2 * e0 + sw] : 0;
      sk1 = val1 ? keys[2 * e1 + sw] : 0;
    } else {
      sk0 = (AOS && !LEAN && sort_word)
          ? val_reg[(AOS && !LEAN) ? i0 : 0] : key_reg[(AOS && LEAN) ? 0 : i0];
      sk1 = (AOS && !LEAN && sort_word)
          ? val_reg[(AOS && !LEAN) ? i1 : 0] : key_reg[(AOS && LEAN) ? 0 : i1];
    }
    uint32_t d0 = (uint32_t)(sk0 >> shift) & (ND - 1);
    uint32_t d1 = (uint32_t)(sk1 >> shift) & (ND - 1);
    uint64_t vm0 = __ballot(val0), vm1 = __ballot(val1);
    uint64_t v0[PBITS], v1[PBITS];
#pragma unroll
    for (int b = 0; b < PBITS; ++b) v0[b] = __ballot((d0 >> b) & 1);
#pragma unroll
    for (int b = 0; b < PBITS; ++b) v1[b] = __ballot((d1 >> b) & 1);
    // per-chunk bases read BEFORE this chunk's single update round
    uint32_t base0 = (val0 && stage != 9) ? my[d0] : 0;
    uint32_t base1 = (val1 && stage != 9) ? my[d1] : 0;
    uint64_t m0 = match_votes<PBITS>(v0, vm0, d0);
    uint64_t m1 = match_votes<PBITS>(v1, vm1, d1);
    uint64_t cross1 = match_votes<PBITS>(v0, vm0, d1);  // i0 elems == d1
    uint32_t r0 = base0 + (uint32_t)__popcll(m0 & lt);
    uint32_t r1 = base1 + (uint32_t)__popcll(cross1)
                  + (uint32_t)__popcll(m1 & lt);
    digrank[i0] = (d0 << 16) | (val0 ? r0 : 0);
    digrank[i1] = (d1 << 16) | (val1 ? r1 : 0);
    if (stage != 9) {
      // one LDS update per digit per PAIR: the i0 leader of a digit adds
      // both iterations' counts; digits present only in i1 are added by
      // their i1 leader
      if (val0 && (m0 & lt) == 0) {
        uint64_t cross0 = match_votes<PBITS>(v1, vm1, d0);
        my[d0] = base0 + (uint32_t)__popcll(m0) + (uint32_t)__popcll(cross0);
      }
      if (val1 && (m1 & lt) == 0 && cross1 == 0)
        my[d1] = base1 + (uint32_t)__popcll(m1);
    }
  }
  __syncthreads();

  // per-digit wave-exclusive scan + block totals into start[]
  for (int d = tid; d < ND; d += BS) {
    uint32_t run = 0;
#pragma unroll
    for (int w = 0; w < NWT; ++w) {
      uint32_t t = counters[w * ND + d];
      counters[w * ND + d] = run;
      run += t;
    }
    start[d] = run;
  }
  __syncthreads();

  // publish this block's AGGREGATE as early as possible — consumers'
  // lookbacks are gated on it; our own walk is DEFERRED to overlap the
  // LDS exchange below (the walk is wait-dominated: it costs ~30% of the
  // pass when run serially here — profiles/r01)
  uint64_t my_total = 0;
  if (hist_pref) {
    // no-lookback mode: cross-block prefixes were precomputed by the
    // hist+scan pre-pass (kills the ~10us/block lookback wait)
    if (tid < ND) pref[tid] = hist_pref[(uint64_t)b * ND + tid];
  } else if (tid < ND) {
    my_total = start[tid];
    uint64_t* slot = lb_mode
        ? &desc[(uint64_t)tid * gridDim.x + b]
        : &desc[(uint64_t)b * ND + tid];
    if (stage == 0 || b == 0) {  // stage 0: ablation, WRONG results
      __hip_atomic_store(slot, FLAG_INC | my_total, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
      pref[tid] = 0;
    } else {
      __hip_atomic_store(slot, FLAG_AGG | my_total, __ATOMIC_RELAXED,
                         __HIP_MEMORY_SCOPE_AGENT);
    }
  }
  // block-local digit starts (exclusive scan of totals)
  block_exscan<BS>(start, sums, ND);  // includes the needed __syncthreads

  uint64_t t1 = timing ? __builtin_amdgcn_s_memrealtime() : 0;
  if (AOS) {
    uint64_t t2 = t1, t3 = t1;
    const uint32_t off_mask = stage < 3 ? 1023u : 0xFFFFFFFFu;
    const int rounds = split_exch > 1 ? split_exch : 1;
    for (int round = 0; round < rounds; ++round) {
      if (round) __syncthreads();     // previous half's writeout done
      if (stage >= 2) {
        // pair exchange first: the deferred lookback's wait overlaps it
#pragma unroll
        for (int i = 0; i < IT; ++i) {
          uint64_t e = chunk + (uint64_t)i * kWave + lane;
          if (e < n) {
            uint32_t d = digrank[i] >> 16;
            uint32_t j = start[d] + my[d] + (digrank[i] & 0xFFFF);
            uint32_t jl = j - (uint32_t)(round * ecap);
            if (jl < (uint32_t)ecap) {
              if (LEAN)
                exch2[jl] = reinterpret_cast<const u64x2*>(keys)[e];
              else
                exch2[jl] = u64x2{key_reg[i],
                                  val_reg[(AOS && !LEAN) ? i : 0]};
            }
          }
        }
      }
      if (round == 0) {
        t2 = timing ? __builtin_amdgcn_s_memrealtime() : 0;
        if (!hist_pref && stage != 0 && b != 0 && tid < ND) {
          if (lb_mode)
            lookback_walk_t<ND>(desc, gridDim.x, b, tid, my_total, pref);
          else
            lookback_walk<ND>(desc, b, tid, my_total, pref);
        }
        if (stage < 2 && !timing) return;  // ablation
      }
      __syncthreads();
      if (round == 0) t3 = timing ? __builtin_amdgcn_s_memrealtime() : 0;
      if (FUSE_GATHER) {
        // copy each slot's RECORD to its final position: key_dst[d]
        // holds per-digit record bases; word-mapped like gather_records
        // (writes coalesce within digit runs, reads are the random
        // per-record chunks the separate gather paid anyway)
        const uint32_t total_w = tile_n * rec_w4;
        for (uint32_t w = tid; w < total_w; w += BS) {
          uint32_t j = w / rec_w4, o = w - j * rec_w4;
          u64x2 kv = exch2[j];
          uint32_t d = (uint32_t)((uint64_t)(sort_word ? kv.y : kv.x)
                                  >> shift) & (ND - 1);
          uint32_t off = (pref[d] + (j - start[d])) & off_mask;
          uint64_t idx = (uint64_t)kv.y & ((1ull << 48) - 1);
          reinterpret_cast<uint32_t*>(key_dst[d])[(uint64_t)off * rec_w4 + o] =
              rec_words[idx * rec_w4 + o];
        }
      } else {
#pragma unroll
      for (int i = 0; i < IT; ++i) {
        uint32_t jl = (uint32_t)i * BS + tid;
        if (jl >= (uint32_t)ecap) break;
        uint32_t j = (uint32_t)round * ecap + jl;
        if (j < tile_n) {
          u64x2 kv = exch2[jl];
          uint32_t d = (uint32_t)((uint64_t)(sort_word ? kv.y : kv.x)
                                  >> shift) & (ND - 1);
          uint32_t off = (pref[d] + (j - start[d])) & off_mask;
          reinterpret_cast<u64x2*>(key_dst[d])[off] = kv;
        }
      }
      }
    }
    if (timing && tid == 0) {
      uint64_t t4 = __builtin_amdgcn_s_memrealtime();
      atomicAdd(&timing[0], t1 - t0);  // phase A (+scans+publish)
      atomicAdd(&timing[1], t2 - t1);  // exchange
      atomicAdd(&timing[2], t3 - t2);  // lookback walk + barrier
      atomicAdd(&timing[3], t4 - t3);  // writeout (+2nd round when split)
    }
    return;
  }
  if (!hist_pref && b != 0 && tid < ND) {
    if (lb_mode)
      lookback_walk_t<ND>(desc, gridDim.x, b, tid, my_total, pref);
    else
      lookback_walk<ND>(desc, b, tid, my_total, pref);
  }

  // SoA path: key exchange + write-out, then val exchange + write-out
#pragma unroll
  for (int i = 0; i < IT; ++i) {
    uint64_t e = chunk + (uint64_t)i * kWave + lane;
    if (e < n) {
      uint32_t d = digrank[i] >> 16;
      uint32_t j = start[d] + my[d] + (digrank[i] & 0xFFFF);
      digrank[i] = j;  // reuse: local sorted position
      exch[j] = key_reg[i];
    }
  }
  __syncthreads();
#pragma unroll
  for (int i = 0; i < IT; ++i) {
    uint32_t j = i * BS + tid;
    if (j < tile_n) {
      uint64_t k = exch[j];
      uint32_t d = (uint32_t)(k >> shift) & (ND - 1);
      uint32_t off = pref[d] + (j - start[d]);
      dsort[j] = (uint8_t)d;
      reinterpret_cast<uint64_t*>(key_dst[d])[off] = k;
    }
  }
  if (HAS_VAL) {
    __syncthreads();
#pragma unroll 1
    for (int i = 0; i < IT; ++i) {
      uint64_t e = chunk + (uint64_t)i * kWave + lane;
      if (e < n) exch[digrank[i]] = vals[e];
    }
    __syncthreads();
#pragma unroll 1
    for (int i = 0; i < IT; ++i) {
      uint32_t j = i * BS + tid;
      if (j < tile_n) {
        uint32_t d = dsort[j];
        uint32_t off = pref[d] + (j - start[d]);
        reinterpret_cast<uint64_t*>(val_dst[d])[off] = exch[j];
      }
    }
  }
}

// digit bases for pass p from the all-pass totals; rec_bytes = 8 (SoA)
// or 16 (AoS interleaved records)
template <int ND = 256>
__global__ void onesweep_digit_bases_kernel(
    const uint32_t* __restrict__ totals /* [pass][ND] */, int pass,
    uint64_t out_keys, uint64_t out_vals, uint64_t* __restrict__ key_dst,
    uint64_t* __restrict__ val_dst, int rec_bytes) {
  __shared__ uint32_t arr[ND];
  __shared__ uint32_t sums[BLOCK];
  const int tid = threadIdx.x;
  if (tid < ND) arr[tid] = totals[pass * ND + tid];
  __syncthreads();
  block_exscan(arr, sums, ND);
  if (tid < ND) {
    key_dst[tid] = out_keys + (uint64_t)arr[tid] * rec_bytes;
    val_dst[tid] = out_vals ? out_vals + (uint64_t)arr[tid] * rec_bytes : 0;
  }
}

// ---------------------------------------------------------------------------
// Wide-record machinery (VERDICT r01 missing item 3: the reference serves
// WHATEVER bytes the writer produced — RdmaMappedFile.java:113-157 —
// while the r01 fast path handled only 16-byte records; canonical
// TeraSort is 10 B key + 90 B value = 100 B records).
//
// Design: records stay put while 16-byte (sortkey, aux) PAIRS flow
// through the existing radix machinery, then ONE gather pass moves each
// W-byte record to its final position:
//   * extract_pairs: record -> (key-prefix u64, aux = keylo16<<48 | idx)
//   * partition/sort the pairs (existing kernels; `key_word` selects
//     which u64 of the pair the digit comes from, so the 80-bit key
//     sorts as 16 aux bits then 64 prefix bits, LSD-stable)
//   * gather_records: walk the grouped/sorted pairs linearly, copy record
//     idx to its slot — writes are linear (W-byte runs), reads are
//     W-byte contiguous chunks; both sides move each record exactly once.
// Traffic: ~2.2x the record bytes + ~64 B/pass for the 16 B pairs —
// pair passes cost 16% of what full-record passes would at W = 100.

// Key layout inside a record: bytes [0,8) u64 LE prefix, bytes [8,10)
// u16 LE low bits (key_bytes == 8 or 10); 80-bit order = (prefix, lo).
// pid_func >= 0: pair.x carries the PARTITION ID of the record instead
// of the raw prefix — the two-level >4096-partition path then radixes
// the pid itself (coarse bits then fine bits).
__global__ __launch_bounds__(BLOCK) void extract_pairs_kernel(
    const uint8_t* __restrict__ recs, uint64_t n, uint32_t rec_bytes,
    uint32_t key_bytes, uint64_t* __restrict__ pairs, int pid_func,
    int pid_shift, uint32_t pid_mask, uint32_t pid_nparts,
    uint64_t idx_base /* aux indices are idx_base + e: per-chunk extracts
                         reference records in a larger arena */) {
  const uint64_t stride = (uint64_t)gridDim.x * BLOCK;
  for (uint64_t e = (uint64_t)blockIdx.x * BLOCK + threadIdx.x; e < n;
       e += stride) {
    const uint8_t* p = recs + e * rec_bytes;
    // 4-byte-aligned loads (rec_bytes % 4 == 0 enforced host-side)
    uint32_t lo = *reinterpret_cast<const uint32_t*>(p);
    uint32_t hi = *reinterpret_cast<const uint32_t*>(p + 4);
    uint64_t prefix = ((uint64_t)hi << 32) | lo;
    uint64_t aux = idx_base + e;
    if (key_bytes > 8) {
      uint16_t klo = (uint16_t)(*reinterpret_cast<const uint32_t*>(p + 8));
      aux |= (uint64_t)klo << 48;
    }
    uint64_t x = pid_func < 0
        ? prefix
        : (uint64_t)digit_of(prefix, pid_shift, pid_mask, pid_func,
                             pid_nparts);
    reinterpret_cast<ulonglong2*>(pairs)[e] = ulonglong2{x, aux};
  }
}

// Move records to their final slots following the grouped/sorted pairs.
// Slot j's record index comes from pairs[2j+1]; its destination is
//   dst_mode 0: out_base + j*rec_bytes                  (sorted output)
//   dst_mode 1: dst_addr[d] + (j - dstart[d])*rec_bytes (partition path,
//               d = digit_of(pairs[2j]) — per-digit HBM block positions)
constexpr uint64_t AUX_IDX_MASK = (1ull << 48) - 1;

__global__ __launch_bounds__(BLOCK) void gather_records_kernel(
    const uint32_t* __restrict__ recs, const uint64_t* __restrict__ pairs,
    uint64_t n, uint32_t w4 /* rec_bytes/4 */, uint32_t recs_per_block,
    int dst_mode, uint64_t out_base,
    const uint64_t* __restrict__ dst_addr,
    const uint32_t* __restrict__ dstart, int shift, uint32_t mask,
    int func, uint32_t nparts) {
  const uint64_t r0 = (uint64_t)blockIdx.x * recs_per_block;
  if (r0 >= n) return;
  const uint32_t nr = (uint32_t)min((uint64_t)recs_per_block, n - r0);
  const uint32_t total = nr * w4;
  for (uint32_t w = threadIdx.x; w < total; w += BLOCK) {
    uint32_t r = w / w4, o = w - r * w4;
    uint64_t j = r0 + r;
    ulonglong2 pr = reinterpret_cast<const ulonglong2*>(pairs)[j];
    uint64_t idx = pr.y & AUX_IDX_MASK;
    uint32_t* dst;
    if (dst_mode == 0) {
      dst = reinterpret_cast<uint32_t*>(out_base) + j * w4;
    } else {
      uint32_t d = digit_of(pr.x, shift, mask, func, nparts);
      dst = reinterpret_cast<uint32_t*>(dst_addr[d]) +
            (uint64_t)(uint32_t)(j - dstart[d]) * w4;
    }
    // plain cached accesses: streaming (nontemporal) hints measured
    // 252 vs 256 GB/s on the flagship — the L2 reuse between the
    // producing kernels and this gather outweighs the pollution
    dst[o] = recs[idx * w4 + o];
  }
}

void extract_pairs(uintptr_t recs, uint64_t n, uint32_t rec_bytes,
                   uint32_t key_bytes, uintptr_t pairs, uintptr_t stream,
                   int pid_func, int pid_shift, uint32_t pid_mask,
                   uint32_t pid_nparts, uint64_t idx_base) {
  if (rec_bytes % 4 || rec_bytes < 8)
    throw std::runtime_error("rec_bytes must be a multiple of 4, >= 8");
  if (key_bytes != 8 && key_bytes != 10)
    throw std::runtime_error("key_bytes must be 8 or 10");
  if (n >> 48)
    throw std::runtime_error("record count exceeds 2^48");
  auto s = reinterpret_cast<hipStream_t>(stream);
  uint32_t grid = (uint32_t)min((n + BLOCK - 1) / BLOCK, (uint64_t)4096);
  if ((idx_base + n) >> 48)
    throw std::runtime_error("idx_base + n exceeds 2^48");
  hipLaunchKernelGGL(extract_pairs_kernel, dim3(grid ? grid : 1),
                     dim3(BLOCK), 0, s,
                     reinterpret_cast<const uint8_t*>(recs), n, rec_bytes,
                     key_bytes, reinterpret_cast<uint64_t*>(pairs),
                     pid_func, pid_shift, pid_mask, pid_nparts, idx_base);
  HIP_CHECK(hipGetLastError());
}

void gather_records(uintptr_t recs, uintptr_t pairs, uint64_t n,
                    uint32_t rec_bytes, int dst_mode, uint64_t out_base,
                    uintptr_t dst_addr, uintptr_t dstart, int shift,
                    uint32_t mask, int func, uint32_t nparts,
                    uintptr_t stream) {
  if (rec_bytes % 4) throw std::runtime_error("rec_bytes % 4 != 0");
  auto s = reinterpret_cast<hipStream_t>(stream);
  uint32_t w4 = rec_bytes / 4;
  uint32_t rpb = 16384 / w4;           // ~16k words per block
  if (rpb < 1) rpb = 1;
  uint64_t grid = (n + rpb - 1) / rpb;
  if (grid > 0x7FFFFFFF) throw std::runtime_error("grid too large");
  hipLaunchKernelGGL(gather_records_kernel, dim3((uint32_t)(grid ? grid : 1)),
                     dim3(BLOCK), 0, s,
                     reinterpret_cast<const uint32_t*>(recs),
                     reinterpret_cast<const uint64_t*>(pairs), n, w4, rpb,
                     dst_mode, out_base,
                     reinterpret_cast<const uint64_t*>(dst_addr),
                     reinterpret_cast<const uint32_t*>(dstart), shift, mask,
                     func, nparts);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Microbench probe: emulate the scatter's write pattern — each block
// writes its tile as `nregions` bursts of `burst` bytes at region-strided
// scattered offsets. Measures HBM efficiency vs burst length to size the
// optimization headroom (not part of the data path).

__global__ __launch_bounds__(BLOCK) void probe_scatter_write_kernel(
    uint64_t* __restrict__ out, uint64_t region_elems, uint32_t nregions,
    uint32_t burst_elems, uint32_t bursts_per_block) {
  // block b writes bursts_per_block bursts; burst i goes to region
  // (b*7 + i) % nregions at offset (b * bursts_per_block + i) scaled —
  // scattered across regions like the radix scatter's digit streams.
  const uint32_t b = blockIdx.x;
  for (uint32_t i = 0; i < bursts_per_block; ++i) {
    uint32_t region = (b * 7u + i) % nregions;
    uint64_t off_in_region =
        ((uint64_t)(b * 1315423911u + i * 2654435761u)) %
        (region_elems - burst_elems);
    uint64_t* dst = out + (uint64_t)region * region_elems + off_in_region;
    for (uint32_t e = threadIdx.x; e < burst_elems; e += BLOCK)
      dst[e] = (uint64_t)b * i + e;
  }
}

void probe_scatter_write(uintptr_t out, uint64_t region_elems,
                         uint32_t nregions, uint32_t burst_elems,
                         uint32_t bursts_per_block, uint32_t grid,
                         uintptr_t stream) {
  hipLaunchKernelGGL(probe_scatter_write_kernel, dim3(grid), dim3(BLOCK), 0,
                     reinterpret_cast<hipStream_t>(stream),
                     reinterpret_cast<uint64_t*>(out), region_elems, nregions,
                     burst_elems, bursts_per_block);
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// Sort-merge join (merge kernel for SQL sort-merge join).
// Both sides radix-sorted first; the merge is per-A-row binary search in
// sorted B (log nb coherent loads — neighbouring threads probe
// neighbouring B ranges, so the walk stays in L2), then pair emission at
// scanned offsets.

__device__ __forceinline__ uint32_t lower_bound_u64(
    const uint64_t* __restrict__ arr, uint32_t n, uint64_t key) {
  uint32_t lo = 0, hi = n;
  while (lo < hi) {
    uint32_t mid = (lo + hi) >> 1;
    if (arr[mid] < key) lo = mid + 1; else hi = mid;
  }
  return lo;
}

__global__ __launch_bounds__(BLOCK) void join_count_kernel(
    const uint64_t* __restrict__ a_keys, uint32_t na,
    const uint64_t* __restrict__ b_keys, uint32_t nb_,
    uint32_t* __restrict__ counts, uint32_t* __restrict__ lo_idx) {
  const uint64_t stride = (uint64_t)gridDim.x * BLOCK;
  for (uint64_t i = (uint64_t)blockIdx.x * BLOCK + threadIdx.x; i < na;
       i += stride) {
    uint64_t k = a_keys[i];
    uint32_t lo = lower_bound_u64(b_keys, nb_, k);
    uint32_t hi = lo;
    while (hi < nb_ && b_keys[hi] == k) ++hi;  // runs are short for ~unique keys
    counts[i] = hi - lo;
    lo_idx[i] = lo;
  }
}

__global__ __launch_bounds__(BLOCK) void join_emit_kernel(
    const uint64_t* __restrict__ a_keys, const uint64_t* __restrict__ a_vals,
    uint32_t na, const uint64_t* __restrict__ b_vals,
    const uint32_t* __restrict__ counts, const uint32_t* __restrict__ lo_idx,
    const uint64_t* __restrict__ offsets,
    uint64_t* __restrict__ out_key, uint64_t* __restrict__ out_a,
    uint64_t* __restrict__ out_b) {
  const uint64_t stride = (uint64_t)gridDim.x * BLOCK;
  for (uint64_t i = (uint64_t)blockIdx.x * BLOCK + threadIdx.x; i < na;
       i += stride) {
    uint32_t cnt = counts[i];
    if (!cnt) continue;
    uint64_t off = offsets[i];
    uint32_t lo = lo_idx[i];
    uint64_t k = a_keys[i];
    uint64_t av = a_vals ? a_vals[i] : 0;
    for (uint32_t j = 0; j < cnt; ++j) {
      if (out_key) out_key[off + j] = k;
      if (out_a) out_a[off + j] = av;
      out_b[off + j] = b_vals[lo + j];
    }
  }
}

static inline uint32_t join_grid(uint32_t n) {
  uint32_t g = (n + BLOCK - 1) / BLOCK;
  return g > 2048 ? 2048 : (g ? g : 1);
}

void join_count(uintptr_t a_keys, uint32_t na, uintptr_t b_keys, uint32_t nb_,
                uintptr_t counts, uintptr_t lo_idx, uintptr_t stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(join_count_kernel, dim3(join_grid(na)), dim3(BLOCK), 0,
                     s, reinterpret_cast<const uint64_t*>(a_keys), na,
                     reinterpret_cast<const uint64_t*>(b_keys), nb_,
                     reinterpret_cast<uint32_t*>(counts),
                     reinterpret_cast<uint32_t*>(lo_idx));
  HIP_CHECK(hipGetLastError());
}

void join_emit(uintptr_t a_keys, uintptr_t a_vals, uint32_t na,
               uintptr_t b_vals, uintptr_t counts, uintptr_t lo_idx,
               uintptr_t offsets, uintptr_t out_key, uintptr_t out_a,
               uintptr_t out_b, uintptr_t stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  hipLaunchKernelGGL(join_emit_kernel, dim3(join_grid(na)), dim3(BLOCK), 0, s,
                     reinterpret_cast<const uint64_t*>(a_keys),
                     reinterpret_cast<const uint64_t*>(a_vals), na,
                     reinterpret_cast<const uint64_t*>(b_vals),
                     reinterpret_cast<const uint32_t*>(counts),
                     reinterpret_cast<const uint32_t*>(lo_idx),
                     reinterpret_cast<const uint64_t*>(offsets),
                     reinterpret_cast<uint64_t*>(out_key),
                     reinterpret_cast<uint64_t*>(out_a),
                     reinterpret_cast<uint64_t*>(out_b));
  HIP_CHECK(hipGetLastError());
}

// ---------------------------------------------------------------------------
// host wrappers

static inline uint32_t num_tiles(uint32_t n) {
  return (uint32_t)(((uint64_t)n + TILE - 1) / TILE);
}

size_t radix_hist_bytes(uint32_t n, int nbits) {
  return (size_t)(1u << nbits) * num_tiles(n) * sizeof(uint32_t);
}

template <int NBITS>
static void hist_launch(const uint64_t* keys, uint32_t n, int shift,
                        uint32_t* hist, hipStream_t s, int func = 0,
                        int in_stride = 1, uint32_t nparts = 0) {
  uint32_t nb = num_tiles(n);
  size_t lds = (size_t)NW * (1 << NBITS) * 4;
  hipLaunchKernelGGL(radix_hist_kernel<NBITS>, dim3(nb), dim3(BLOCK), lds, s,
                     keys, n, shift, hist, nb, func, in_stride, nparts);
  HIP_CHECK(hipGetLastError());
}

template <int NBITS, bool HAS_VAL>
static void scatter_launch(const uint64_t* keys, const uint64_t* vals,
                           uint32_t n, int shift, const uint32_t* hist,
                           const uint64_t* key_dst, const uint64_t* val_dst,
                           hipStream_t s, int func = 0, int aos_out = 0,
                           int in_stride = 1, uint32_t nparts = 0) {
  constexpr int ND = 1 << NBITS;
  uint32_t nb = num_tiles(n);
  size_t lds = (size_t)TILE * 8 + (size_t)NW * ND * 4 + (size_t)ND * 4 * 2 +
               BLOCK * 4;
  auto kfn = radix_scatter_kernel<NBITS, HAS_VAL>;
  static bool attr_set[13] = {};
  if (lds > 64 * 1024 && !attr_set[NBITS]) {
    (void)hipFuncSetAttribute(
        reinterpret_cast<const void*>(kfn),
        hipFuncAttributeMaxDynamicSharedMemorySize, (int)lds);
    attr_set[NBITS] = true;
  }
  hipLaunchKernelGGL(kfn, dim3(nb), dim3(BLOCK), lds, s, keys, vals, n, shift,
                     hist, nb, key_dst, val_dst, func, aos_out, in_stride,
                     nparts);
  HIP_CHECK(hipGetLastError());
}

#define DISPATCH_NBITS(nbits, FN, ...)                        \
  switch (nbits) {                                            \
    case 4: FN<4>(__VA_ARGS__); break;                        \
    case 5: FN<5>(__VA_ARGS__); break;                        \
    case 6: FN<6>(__VA_ARGS__); break;                        \
    case 7: FN<7>(__VA_ARGS__); break;                        \
    case 8: FN<8>(__VA_ARGS__); break;                        \
    case 9: FN<9>(__VA_ARGS__); break;                        \
    case 10: FN<10>(__VA_ARGS__); break;                      \
    case 11: FN<11>(__VA_ARGS__); break;                      \
    case 12: FN<12>(__VA_ARGS__); break;                      \
    default: throw std::runtime_error("nbits must be 4..12"); \
  }

void radix_hist(uintptr_t keys, uint32_t n, int shift, int nbits,
                uintptr_t hist, uintptr_t stream, int func, int in_stride,
                uint32_t nparts) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  DISPATCH_NBITS(nbits, hist_launch, reinterpret_cast<const uint64_t*>(keys),
                 n, shift, reinterpret_cast<uint32_t*>(hist), s, func,
                 in_stride, nparts);
}

size_t radix_scan_ws_bytes(uint32_t n, int nbits) {
  uint32_t nb = num_tiles(n);
  uint32_t chunk = scan_chunk_rows(nb);
  uint32_t chunks = (nb + chunk - 1) / chunk;
  return (size_t)chunks * (1u << nbits) * sizeof(uint32_t);
}

template <int NBITS>
static void scan_launch(uint32_t* hist, uint32_t nb, uint32_t* partial,
                        uint32_t* totals, hipStream_t s) {
  constexpr int ND = 1 << NBITS;
  uint32_t chunk = scan_chunk_rows(nb);
  uint32_t chunks = (nb + chunk - 1) / chunk;
  hipLaunchKernelGGL(scan_chunk_sums_kernel<ND>, dim3(chunks), dim3(BLOCK), 0,
                     s, hist, nb, chunk, partial);
  HIP_CHECK(hipGetLastError());
  hipLaunchKernelGGL(scan_chunks_kernel<ND>, dim3(1), dim3(BLOCK), 0, s,
                     partial, chunks, totals);
  HIP_CHECK(hipGetLastError());
  hipLaunchKernelGGL(scan_rewrite_kernel<ND>, dim3(chunks), dim3(BLOCK), 0, s,
                     hist, nb, chunk, partial);
  HIP_CHECK(hipGetLastError());
}

void radix_scan(uintptr_t hist, uint32_t n, int nbits, uintptr_t totals,
                uintptr_t scan_ws, uintptr_t stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  uint32_t nb = num_tiles(n);
  DISPATCH_NBITS(nbits, scan_launch, reinterpret_cast<uint32_t*>(hist), nb,
                 reinterpret_cast<uint32_t*>(scan_ws),
                 reinterpret_cast<uint32_t*>(totals), s);
}

template <int NBITS>
static void scatter_hv(const uint64_t* keys, const uint64_t* vals, uint32_t n,
                       int shift, const uint32_t* hist, const uint64_t* kd,
                       const uint64_t* vd, hipStream_t s, int func = 0,
                       int aos_out = 0, int in_stride = 1,
                       uint32_t nparts = 0) {
  if (vals)
    scatter_launch<NBITS, true>(keys, vals, n, shift, hist, kd, vd, s,
                                func, aos_out, in_stride, nparts);
  else
    scatter_launch<NBITS, false>(keys, nullptr, n, shift, hist, kd, nullptr,
                                 s, func, aos_out, in_stride, nparts);
}

void radix_scatter(uintptr_t keys, uintptr_t vals, uint32_t n, int shift,
                   int nbits, uintptr_t hist, uintptr_t key_dst,
                   uintptr_t val_dst, uintptr_t stream, int func,
                   int aos_out, int in_stride, uint32_t nparts) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  DISPATCH_NBITS(nbits, scatter_hv, reinterpret_cast<const uint64_t*>(keys),
                 reinterpret_cast<const uint64_t*>(vals), n, shift,
                 reinterpret_cast<const uint32_t*>(hist),
                 reinterpret_cast<const uint64_t*>(key_dst),
                 reinterpret_cast<const uint64_t*>(val_dst), s, func,
                 aos_out, in_stride, nparts);
}

// Full LSD sort of (keys[, vals]) by bits [start_bit, end_bit).
// ws layout: hist u32[ND*nb] | totals u32[256] | key_dst u64[256] |
//            val_dst u64[256].  Returns which buffer holds the result:
// 0 = keys/vals, 1 = tmp_keys/tmp_vals.
size_t sort_workspace_bytes(uint32_t n) {
  return radix_hist_bytes(n, 8) + radix_scan_ws_bytes(n, 8) + 256 * 4 +
         256 * 8 * 2;
}

// Onesweep sort: ws layout = totals u32[passes*256] | key_dst u64[256] |
// val_dst u64[256] | ticket u32 (+pad) | desc u32[nb*256].
constexpr int OS_ITEMS = 16;  // SoA: 4096-elem tiles @ 256 threads
// AoS tile selected at runtime: 4096 (512thr x IT8, 2 blocks/CU) or
// 8192 (512thr x IT16, 1 block/CU, 512B write bursts)
static int g_aos_tile = 4096;
void set_aos_tile(int t) { g_aos_tile = t; }
// ablation: 1 = rank+lookback only, 2 = full work but stores land in a
// compact dummy window (isolates scattered-store cost), 3 = normal
static int g_pass_stage = 3;
void set_pass_stage(int s) { g_pass_stage = s; }
// sort cross-block-prefix mode: 0 = decoupled lookback (one kernel per
// pass; measured fastest: 6.2 vs 7.0 ms for the hist+scan pre-pass at
// 64M — the pre-pass re-read outweighs the lookback wait), 1 = hist+scan
static int g_sort_mode = 0;
void set_sort_mode(int m) { g_sort_mode = m; }
// optional phase-timing accumulator: u64[4] = {phaseA, exch, lookback, writeout}
static uint64_t* g_timing_buf = nullptr;
void set_timing_buf(uintptr_t p) { g_timing_buf = reinterpret_cast<uint64_t*>(p); }
// descriptor layout: 0 = [nb][ND] (r01), 1 = transposed [ND][nb] (walk
// reads contiguous backwards — 16 descriptors/line)
static int g_lb_mode = 0;
void set_lookback_mode(int m) { g_lb_mode = m; }
// AoS split exchange: halve the LDS exchange buffer (2 rounds) so 3
// blocks/CU stay resident and hide more of the lookback wait
static int g_split_exch = 0;
void set_split_exchange(int m) { g_split_exch = m; }
// AoS register-lean variant (re-reads pairs at exchange): 3 blocks/CU
static int g_lean = 0;
void set_lean_pass(int m) { g_lean = m; }

static inline uint32_t os_num_tiles_t(uint32_t n, int tile) {
  return (uint32_t)(((uint64_t)n + tile - 1) / tile);
}
static inline uint32_t os_num_tiles(uint32_t n) {
  return os_num_tiles_t(n, BLOCK * OS_ITEMS);
}

size_t onesweep_workspace_bytes(uint32_t n, int passes) {
  uint32_t nb = os_num_tiles_t(n, 4096);  // smallest tile = max desc size
  return (size_t)passes * 256 * 4 + 256 * 8 * 2 + 16 +
         (size_t)nb * 256 * 8;
}

// aos = 0: keys/vals are separate u64 arrays (SoA).
// aos = 1: keys/tmp_keys point at interleaved (key,val) 16-byte records;
//          vals/tmp_vals ignored.
// PBITS: digit width per pass. 7-bit digits double the per-digit write
// burst (512 B at tile 4096) at the cost of more passes — the caller
// picks per total bit count.
template <int PBITS>
static int onesweep_sort_tmpl(uintptr_t keys, uintptr_t vals,
                              uintptr_t tmp_keys, uintptr_t tmp_vals,
                              uint32_t n, int start_bit, int end_bit,
                              uintptr_t ws, hipStream_t s, int aos,
                              int sort_word = 0, uintptr_t fuse_recs = 0,
                              uintptr_t fuse_out = 0,
                              uint32_t fuse_rec_bytes = 0) {
  constexpr int PD = 1 << PBITS;
  const int aos_tile = g_aos_tile;
  uint32_t nb = aos ? os_num_tiles_t(n, aos_tile) : os_num_tiles(n);
  int passes = (end_bit - start_bit + PBITS - 1) / PBITS;
  uint32_t* totals = reinterpret_cast<uint32_t*>(ws);
  uint64_t* key_dst = reinterpret_cast<uint64_t*>(totals + (size_t)passes * PD);
  uint64_t* val_dst = key_dst + PD;
  uint32_t* ticket = reinterpret_cast<uint32_t*>(val_dst + PD);
  uint64_t* desc = reinterpret_cast<uint64_t*>(ticket + 4);
  const bool want_hist = g_sort_mode == 1 && aos && g_aos_tile == 4096 &&
                         sort_word == 0;
  if (!want_hist) {
    HIP_CHECK(hipMemsetAsync(totals, 0, (size_t)passes * PD * 4, s));
    uint32_t hist_grid = nb < 1024 ? (nb ? nb : 1) : 1024;
    hipLaunchKernelGGL((onesweep_hist_all_kernel<10, PBITS>), dim3(hist_grid),
                       dim3(BLOCK), 0, s,
                       reinterpret_cast<const uint64_t*>(keys), n, start_bit,
                       passes, totals, aos ? 2 : 1, sort_word);
    HIP_CHECK(hipGetLastError());
  }
  size_t lds_soa = (size_t)BLOCK * OS_ITEMS * 8 + (size_t)NW * PD * 4 +
                   PD * 4 * 2 + BLOCK * 4 + 16 + BLOCK * OS_ITEMS;
  const int se = g_split_exch > 1 ? g_split_exch : 1;
  size_t lds_aos = (size_t)aos_tile * 16 / se +
                   (size_t)(512 / kWave) * PD * 4 + PD * 4 * 2 + 512 * 4 + 16;
  size_t lds = aos ? lds_aos : lds_soa;
  static bool attr_set = false;
  if (!attr_set) {
    for (const void* f :
         {reinterpret_cast<const void*>(
              &onesweep_pass_kernel<true, OS_ITEMS, false, BLOCK, PBITS>),
          reinterpret_cast<const void*>(
              &onesweep_pass_kernel<false, OS_ITEMS, false, BLOCK, PBITS>),
          reinterpret_cast<const void*>(
              &onesweep_pass_kernel<true, 4, true, 512, PBITS>),
          reinterpret_cast<const void*>(
              &onesweep_pass_kernel<true, 8, true, 512, PBITS>),
          reinterpret_cast<const void*>(
              &onesweep_pass_kernel<true, 16, true, 512, PBITS>),
          reinterpret_cast<const void*>(
              &onesweep_pass_kernel<true, 8, true, 512, PBITS, true>),
          reinterpret_cast<const void*>(
              &onesweep_pass_kernel<true, 8, true, 512, PBITS, false,
                                    true>)})
      (void)hipFuncSetAttribute(f, hipFuncAttributeMaxDynamicSharedMemorySize,
                                160 * 1024 - 1024);
    attr_set = true;
  }
  // hist mode: reuse the desc area as [hist u32[nb*PD] | scan partials]
  const bool use_hist = want_hist;
  uint32_t* hist32 = reinterpret_cast<uint32_t*>(desc);
  uint32_t* hist_scan_ws = hist32 + (size_t)nb * PD;
  uintptr_t src_k = keys, src_v = vals, dst_k = tmp_keys, dst_v = tmp_vals;
  int cur = 0;
  for (int p = 0; p < passes; ++p) {
    int sb = start_bit + p * PBITS;
    if (sb > 64 - PBITS) sb = 64 - PBITS;  // same clamp as hist_all
    // fused final pass: the writeout copies whole records to fuse_out
    // (the gather folded into the sort — key_dst becomes record bases)
    const bool fuse = fuse_recs && p == passes - 1 && aos &&
                      aos_tile == 4096 && g_split_exch <= 1 && !g_lean;
    if (use_hist) {
      hist_launch<PBITS>(reinterpret_cast<const uint64_t*>(src_k), n, sb,
                         hist32, s, 0, 2);
      scan_launch<PBITS>(hist32, nb, hist_scan_ws, totals + (size_t)p * PD,
                         s);
    }
    hipLaunchKernelGGL((onesweep_digit_bases_kernel<PD>), dim3(1),
                       dim3(BLOCK), 0, s, totals, p,
                       fuse ? (uint64_t)fuse_out : (uint64_t)dst_k,
                       fuse ? 0 : (uint64_t)dst_v, key_dst, val_dst,
                       fuse ? (int)fuse_rec_bytes : (aos ? 16 : 8));
    HIP_CHECK(hipGetLastError());
    if (!use_hist) {
      HIP_CHECK(hipMemsetAsync(ticket, 0, 16, s));
      HIP_CHECK(hipMemsetAsync(desc, 0, (size_t)nb * PD * 8, s));
    }
    const uint32_t* pass_pref = use_hist ? hist32 : nullptr;
    if (fuse) {   // (ticket/desc were reset in the !use_hist block)
      hipLaunchKernelGGL(
          (onesweep_pass_kernel<true, 8, true, 512, PBITS, false, true>),
          dim3(nb), dim3(512), lds, s,
          reinterpret_cast<const uint64_t*>(src_k), nullptr, n, sb, desc,
          ticket, key_dst, val_dst, g_pass_stage, nullptr, nullptr,
          sort_word, g_lb_mode, 0,
          reinterpret_cast<const uint32_t*>(fuse_recs),
          fuse_rec_bytes / 4);
      HIP_CHECK(hipGetLastError());
      return cur;  // records land in fuse_out; pair buffers are dead
    } else if (aos && aos_tile == 8192) {
      hipLaunchKernelGGL((onesweep_pass_kernel<true, 16, true, 512, PBITS>),
                         dim3(nb), dim3(512), lds, s,
                         reinterpret_cast<const uint64_t*>(src_k), nullptr, n,
                         sb, desc, ticket, key_dst, val_dst, g_pass_stage,
                         g_timing_buf, pass_pref, sort_word, g_lb_mode,
                         g_split_exch);
    } else if (aos && aos_tile == 2048) {
      hipLaunchKernelGGL((onesweep_pass_kernel<true, 4, true, 512, PBITS>),
                         dim3(nb), dim3(512), lds, s,
                         reinterpret_cast<const uint64_t*>(src_k), nullptr, n,
                         sb, desc, ticket, key_dst, val_dst, g_pass_stage,
                         g_timing_buf, pass_pref, sort_word, g_lb_mode,
                         g_split_exch);
    } else if (aos && g_lean) {
      hipLaunchKernelGGL(
          (onesweep_pass_kernel<true, 8, true, 512, PBITS, true>),
          dim3(nb), dim3(512), lds, s,
          reinterpret_cast<const uint64_t*>(src_k), nullptr, n,
          sb, desc, ticket, key_dst, val_dst, g_pass_stage,
          g_timing_buf, pass_pref, sort_word, g_lb_mode, g_split_exch);
    } else if (aos) {
      hipLaunchKernelGGL((onesweep_pass_kernel<true, 8, true, 512, PBITS>),
                         dim3(nb), dim3(512), lds, s,
                         reinterpret_cast<const uint64_t*>(src_k), nullptr, n,
                         sb, desc, ticket, key_dst, val_dst, g_pass_stage,
                         g_timing_buf, pass_pref, sort_word, g_lb_mode,
                         g_split_exch);
    } else if (vals) {
      hipLaunchKernelGGL(
          (onesweep_pass_kernel<true, OS_ITEMS, false, BLOCK, PBITS>),
          dim3(nb), dim3(BLOCK), lds, s,
          reinterpret_cast<const uint64_t*>(src_k),
          reinterpret_cast<const uint64_t*>(src_v), n, sb, desc, ticket,
          key_dst, val_dst, 3, nullptr, nullptr, 0, g_lb_mode);
    } else {
      hipLaunchKernelGGL(
          (onesweep_pass_kernel<false, OS_ITEMS, false, BLOCK, PBITS>),
          dim3(nb), dim3(BLOCK), lds, s,
          reinterpret_cast<const uint64_t*>(src_k), nullptr, n, sb, desc,
          ticket, key_dst, val_dst, 3, nullptr, nullptr, 0, g_lb_mode);
    }
    HIP_CHECK(hipGetLastError());
    std::swap(src_k, dst_k);
    std::swap(src_v, dst_v);
    cur ^= 1;
  }
  // a requested fusion that no pass applied (non-default tile/ablation
  // modes) must NOT silently leave `out` unwritten — caller falls back
  if (fuse_recs) return -1;
  return cur;
}

int onesweep_sort_pairs_u64(uintptr_t keys, uintptr_t vals,
                            uintptr_t tmp_keys, uintptr_t tmp_vals,
                            uint32_t n, int start_bit, int end_bit,
                            uintptr_t ws, uintptr_t stream) {
  return onesweep_sort_tmpl<8>(keys, vals, tmp_keys, tmp_vals, n, start_bit,
                               end_bit, ws,
                               reinterpret_cast<hipStream_t>(stream), 0);
}

int onesweep_sort_aos_u64(uintptr_t pairs, uintptr_t tmp_pairs, uint32_t n,
                          int start_bit, int end_bit, uintptr_t ws,
                          uintptr_t stream) {
  return onesweep_sort_tmpl<8>(pairs, 0, tmp_pairs, 0, n, start_bit, end_bit,
                               ws, reinterpret_cast<hipStream_t>(stream), 1);
}

// sort_word = 1: pairs order by their SECOND u64's bits (the aux word of
// the wide-record path — 80-bit keys sort as 16 aux bits then 64 prefix
// bits across two calls, LSD-stable).
int onesweep_sort_aos_word_u64(uintptr_t pairs, uintptr_t tmp_pairs,
                               uint32_t n, int start_bit, int end_bit,
                               uintptr_t ws, uintptr_t stream,
                               int sort_word) {
  return onesweep_sort_tmpl<8>(pairs, 0, tmp_pairs, 0, n, start_bit, end_bit,
                               ws, reinterpret_cast<hipStream_t>(stream), 1,
                               sort_word);
}

// sort + FUSED final gather: records land sorted in `out` (the last
// pass's writeout copies whole records; the separate gather kernel and
// the final pair round trip disappear).
int onesweep_sort_aos_fused_u64(uintptr_t pairs, uintptr_t tmp_pairs,
                                uint32_t n, int start_bit, int end_bit,
                                uintptr_t ws, uintptr_t stream,
                                int sort_word, uintptr_t recs,
                                uintptr_t out, uint32_t rec_bytes) {
  return onesweep_sort_tmpl<8>(pairs, 0, tmp_pairs, 0, n, start_bit, end_bit,
                               ws, reinterpret_cast<hipStream_t>(stream), 1,
                               sort_word, recs, out, rec_bytes);
}

int onesweep_sort_aos7_u64(uintptr_t pairs, uintptr_t tmp_pairs, uint32_t n,
                           int start_bit, int end_bit, uintptr_t ws,
                           uintptr_t stream) {
  return onesweep_sort_tmpl<7>(pairs, 0, tmp_pairs, 0, n, start_bit, end_bit,
                               ws, reinterpret_cast<hipStream_t>(stream), 1);
}

int sort_pairs_u64(uintptr_t keys, uintptr_t vals, uintptr_t tmp_keys,
                   uintptr_t tmp_vals, uint32_t n, int start_bit, int end_bit,
                   uintptr_t ws, uintptr_t stream) {
  auto s = reinterpret_cast<hipStream_t>(stream);
  uint32_t nb = num_tiles(n);
  uint32_t* hist = reinterpret_cast<uint32_t*>(ws);
  uint32_t* scan_ws = hist + (size_t)256 * nb;
  uint32_t* totals = scan_ws + radix_scan_ws_bytes(n, 8) / 4;
  uint64_t* key_dst = reinterpret_cast<uint64_t*>(totals + 256);
  uint64_t* val_dst = key_dst + 256;
  uintptr_t src_k = keys, src_v = vals, dst_k = tmp_keys, dst_v = tmp_vals;
  int cur = 0;
  for (int bit = start_bit; bit < end_bit; bit += 8) {
    hist_launch<8>(reinterpret_cast<const uint64_t*>(src_k), n, bit, hist, s);
    scan_launch<8>(hist, nb, scan_ws, totals, s);
    hipLaunchKernelGGL(radix_digit_bases_kernel<8>, dim3(1), dim3(BLOCK), 0, s,
                       totals, (uint64_t)dst_k, (uint64_t)dst_v, key_dst,
                       val_dst);
    HIP_CHECK(hipGetLastError());
    scatter_hv<8>(reinterpret_cast<const uint64_t*>(src_k),
                  reinterpret_cast<const uint64_t*>(src_v), n, bit, hist,
                  key_dst, val_dst, s);
    std::swap(src_k, dst_k);
    std::swap(src_v, dst_v);
    cur ^= 1;
  }
  return cur;
}

}  // namespace hipshuffle
