"""GPU kernel correctness vs plain CPU/torch oracles (gfx950 only)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def hs():
    from sparkrdma_amd.ops import load
    return load()


def rand_keys(n, seed=0, bits=64):
    rng = np.random.default_rng(seed)
    hi = (1 << bits) - 1
    k = rng.integers(0, hi, n, dtype=np.uint64, endpoint=True)
    return torch.from_numpy(k.view(np.int64)).cuda()


@pytest.mark.parametrize("n", [1, 63, 64, 4096, 4097, 1_000_000])
@pytest.mark.parametrize("nbits", [8, 11])
def test_radix_partition_counts_and_order(hs, n, nbits):
    from sparkrdma_amd.ops.radix import radix_partition
    keys = rand_keys(n, seed=n + nbits)
    vals = keys.clone()  # payload = key for integrity checking
    shift = 64 - nbits
    counts, keys_out, vals_out = radix_partition(keys, vals, nbits)
    torch.cuda.synchronize()
    k_np = keys.cpu().numpy().view(np.uint64)
    digits = (k_np >> shift).astype(np.int64)
    want_counts = np.bincount(digits, minlength=1 << nbits)
    got_counts = counts.cpu().numpy().astype(np.int64)
    assert np.array_equal(got_counts, want_counts)
    # stable partition: output == stable argsort by digit
    order = np.argsort(digits, kind="stable")
    want_keys = k_np[order]
    got_keys = keys_out.cpu().numpy().view(np.uint64)
    assert np.array_equal(got_keys, want_keys)
    assert np.array_equal(vals_out.cpu().numpy(), keys_out.cpu().numpy())


@pytest.mark.parametrize("n", [1, 1000, 4096, 1_000_000, 4_000_000])
@pytest.mark.parametrize("onesweep", [False, True])
def test_sort_pairs_full(hs, n, onesweep):
    from sparkrdma_amd.ops.radix import sort_pairs
    keys = rand_keys(n, seed=n)
    vals = keys.clone()
    k_sorted, v_sorted = sort_pairs(keys, vals, onesweep=onesweep)
    torch.cuda.synchronize()
    got = k_sorted.cpu().numpy().view(np.uint64)
    want = np.sort(rand_keys(n, seed=n).cpu().numpy().view(np.uint64))
    assert np.array_equal(got, want)
    assert np.array_equal(v_sorted.cpu().numpy(), k_sorted.cpu().numpy())


@pytest.mark.parametrize("n", [1000, 4096, 3_000_000])
def test_sort_pairs_aos(hs, n):
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    keys = rand_keys(n, seed=n + 1)
    pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
    out = sort_pairs_aos(pairs)
    torch.cuda.synchronize()
    got_k = out[0::2].cpu().numpy().view(np.uint64)
    got_v = out[1::2].cpu().numpy().view(np.uint64)
    want = np.sort(keys.cpu().numpy().view(np.uint64))
    assert np.array_equal(got_k, want)
    assert np.array_equal(got_v, got_k)


def test_sort_aos_stability(hs):
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    n = 400_000
    rng = np.random.default_rng(4)
    k = (rng.integers(0, 500, n, dtype=np.uint64))
    keys = torch.from_numpy(k.view(np.int64)).cuda()
    idx = torch.arange(n, dtype=torch.int64).cuda()
    pairs = torch.stack([keys, idx], dim=1).reshape(-1).contiguous()
    out = sort_pairs_aos(pairs, 0, 16)
    torch.cuda.synchronize()
    order = np.argsort(k, kind="stable")
    assert np.array_equal(out[1::2].cpu().numpy(), order)


def test_onesweep_matches_3kernel_repeated(hs):
    """Race screen: the lookback path must be deterministic across runs
    and identical to the 3-kernel path."""
    from sparkrdma_amd.ops.radix import sort_pairs
    n = 2_000_000
    keys = rand_keys(n, seed=77)
    vals = rand_keys(n, seed=78)
    ref_k, ref_v = sort_pairs(keys.clone(), vals.clone(), onesweep=False)
    for trial in range(5):
        k, v = sort_pairs(keys.clone(), vals.clone(), onesweep=True)
        torch.cuda.synchronize()
        assert torch.equal(k, ref_k), f"trial {trial}: keys diverge"
        assert torch.equal(v, ref_v), f"trial {trial}: vals diverge"


def test_sort_partial_bits(hs):
    """After top-bits partitioning, sorting only the low bits must yield a
    fully sorted partition — the pass-skipping the reduce side relies on."""
    from sparkrdma_amd.ops.radix import sort_pairs
    n = 100_000
    keys = rand_keys(n, seed=3)
    # force top 8 bits constant (single partition's data)
    keys = (keys & ((1 << 56) - 1)) | (0x42 << 56)
    k_sorted, _ = sort_pairs(keys.clone(), None, start_bit=0, end_bit=56)
    torch.cuda.synchronize()
    got = k_sorted.cpu().numpy().view(np.uint64)
    assert np.array_equal(got, np.sort(keys.cpu().numpy().view(np.uint64)))


def test_sort_nonuniform_distribution(hs):
    from sparkrdma_amd.ops.radix import sort_pairs
    n = 500_000
    rng = np.random.default_rng(9)
    # heavily skewed: zipf-ish over a narrow range + duplicates
    k = rng.zipf(1.3, n).astype(np.uint64) % 1000
    keys = torch.from_numpy(k.view(np.int64)).cuda()
    idx = torch.arange(n, dtype=torch.int64).cuda()  # payload = original index
    k_sorted, v_sorted = sort_pairs(keys.clone(), idx)
    torch.cuda.synchronize()
    got_k = k_sorted.cpu().numpy().view(np.uint64)
    assert np.array_equal(got_k, np.sort(k))
    # stability: payload indices within equal-key runs are increasing
    got_v = v_sorted.cpu().numpy()
    order = np.argsort(k, kind="stable")
    assert np.array_equal(got_v, order)


def test_partition_to_explicit_destinations(hs):
    """Map-side path: scatter straight into caller-chosen HBM addresses."""
    from sparkrdma_amd.ops.radix import radix_partition
    n, nbits = 200_000, 8
    nd = 1 << nbits
    keys = rand_keys(n, seed=11)
    vals = keys.clone()
    shift = 64 - nbits
    k_np = keys.cpu().numpy().view(np.uint64)
    digits = (k_np >> shift).astype(np.int64)
    counts_np = np.bincount(digits, minlength=nd)
    # build a deliberately scrambled layout: partition d lands in a
    # per-partition region of an arena, in reverse-digit order
    arena_k = torch.zeros(n, dtype=torch.int64, device="cuda")
    arena_v = torch.zeros(n, dtype=torch.int64, device="cuda")
    bases = np.concatenate([[0], np.cumsum(counts_np[::-1])[:-1]])[::-1].copy()
    key_dst = torch.from_numpy(
        (arena_k.data_ptr() + bases * 8).astype(np.int64)).cuda()
    val_dst = torch.from_numpy(
        (arena_v.data_ptr() + bases * 8).astype(np.int64)).cuda()
    counts, _, _ = radix_partition(keys, vals, nbits,
                                   key_dst=key_dst, val_dst=val_dst)
    torch.cuda.synchronize()
    got = arena_k.cpu().numpy().view(np.uint64)
    order = np.argsort(digits, kind="stable")
    for d in range(nd):
        seg = got[bases[d]:bases[d] + counts_np[d]]
        want = k_np[digits == d]
        assert np.array_equal(seg, want), f"digit {d}"


def test_slab_alloc_and_copy_roundtrip(hs):
    sid = hs.slab_alloc(1 << 20)
    try:
        base = hs.slab_base(sid)
        handle = hs.slab_handle(sid)
        assert len(handle) == 64
        src = torch.arange(1024, dtype=torch.int64, device="cuda")
        ev = hs.read_batch(0, [base], [src.data_ptr()], [1024 * 8])
        hs.wait_event(ev)
        dst = torch.empty(1024, dtype=torch.int64, device="cuda")
        ev = hs.read_batch(0, [dst.data_ptr()], [base], [1024 * 8])
        hs.wait_event(ev)
        assert torch.equal(dst, src)
    finally:
        hs.slab_free(sid)


def test_ipc_same_device_open(hs):
    """IPC open of our own slab from the same process is not allowed by
    HIP; this validates the handle bytes travel & the error is clean."""
    sid = hs.slab_alloc(1 << 20)
    try:
        h = hs.slab_handle(sid)
        assert isinstance(h, bytes) and len(h) == 64
    finally:
        hs.slab_free(sid)


def test_sort_aos_beyond_2pow30(hs):
    """n > 2^30 records: exercises the u64 flag|count descriptor packing
    and u32 prefix arithmetic at the headline-bench scale (the 40 GB job
    sorts 2.68G records in one call). GPU-side checks only."""
    free, _ = torch.cuda.mem_get_info()
    n = 1_200_000_000
    if free < 3 * (2 * n * 8):
        pytest.skip("needs ~58 GB free HBM")
    i = torch.arange(n, dtype=torch.int64, device="cuda")
    keys = i * 0x9E3779B97F4A7C15
    keys ^= keys >> 31                      # covers negative int64s too
    pairs = torch.empty(2 * n, dtype=torch.int64, device="cuda")
    pairs[0::2] = keys
    pairs[1::2] = keys
    del i
    from sparkrdma_amd.ops.radix import sort_pairs_aos
    out = sort_pairs_aos(pairs, 0, 64)
    torch.cuda.synchronize()
    k = out[0::2]
    v = out[1::2]
    assert torch.equal(k, v), "payload permutation diverged from keys"
    # unsigned sortedness: bias the sign bit, compare as int64
    bias = torch.tensor(-2**63, dtype=torch.int64, device="cuda")
    kb = k ^ bias
    assert bool((kb[1:] >= kb[:-1]).all()), "output not sorted (u64 order)"
    # permutation evidence: wrapping int64 sums match the input multiset
    # (commutative mod 2^64; with sortedness + payload equality this
    # pins the output)
    assert int(k.sum().item()) == int(keys.sum().item())


def test_aos_sorter_persistent_buffers(hs):
    """AosSorter: repeated sorts reuse one workspace/ping-pong pair."""
    from sparkrdma_amd.ops.radix import AosSorter
    n = 500_000
    sorter = AosSorter(n)
    for seed in (1, 2):
        keys = rand_keys(n, seed=seed)
        pairs = torch.stack([keys, keys], dim=1).reshape(-1).contiguous()
        out = sorter.sort_(pairs)
        torch.cuda.synchronize()
        got = out[0::2].cpu().numpy().view(np.uint64)
        assert np.array_equal(got, np.sort(keys.cpu().numpy().view(np.uint64)))
