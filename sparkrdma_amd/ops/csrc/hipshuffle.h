// Public surface of the hipshuffle native library.
#pragma once

#include <cstddef>
#include <cstdint>
#include <string>
#include <vector>

namespace hipshuffle {

// pool.cpp
int slab_alloc_id(size_t bytes);
void slab_free_id(int id);
uintptr_t slab_base_id(int id);
std::string slab_handle_id(int id);
uintptr_t ipc_open(const std::string& handle_bytes);
void ipc_close(uintptr_t ptr);
void enable_peer_access(int peer_device);
uint64_t read_batch_ids(int peer, const std::vector<uintptr_t>& dsts,
                        const std::vector<uintptr_t>& srcs,
                        const std::vector<size_t>& sizes);
bool poll_event(uint64_t id);
void wait_event(uint64_t id);
int64_t tcp_recv_chunks(int fd, uintptr_t out, uint64_t total);
int64_t tcp_send_all(int fd, uintptr_t buf, uint64_t n);
uintptr_t host_alloc_pinned(size_t n);
void host_free_pinned(uintptr_t p);
void memcpy_h2d(uintptr_t dst, uintptr_t src, size_t n);
void memcpy_d2h(uintptr_t dst, uintptr_t src, size_t n);

// kernels.hip
size_t radix_hist_bytes(uint32_t n, int nbits);
size_t radix_scan_ws_bytes(uint32_t n, int nbits);
void radix_hist(uintptr_t keys, uint32_t n, int shift, int nbits,
                uintptr_t hist, uintptr_t stream, int func = 0,
                int in_stride = 1, uint32_t nparts = 0);
void radix_scan(uintptr_t hist, uint32_t n, int nbits, uintptr_t totals,
                uintptr_t scan_ws, uintptr_t stream);
void radix_scatter(uintptr_t keys, uintptr_t vals, uint32_t n, int shift,
                   int nbits, uintptr_t hist, uintptr_t key_dst,
                   uintptr_t val_dst, uintptr_t stream, int func = 0,
                   int aos_out = 0, int in_stride = 1, uint32_t nparts = 0);
void extract_pairs(uintptr_t recs, uint64_t n, uint32_t rec_bytes,
                   uint32_t key_bytes, uintptr_t pairs, uintptr_t stream,
                   int pid_func = -1, int pid_shift = 0,
                   uint32_t pid_mask = 0, uint32_t pid_nparts = 0,
                   uint64_t idx_base = 0);
void gather_records(uintptr_t recs, uintptr_t pairs, uint64_t n,
                    uint32_t rec_bytes, int dst_mode, uint64_t out_base,
                    uintptr_t dst_addr, uintptr_t dstart, int shift,
                    uint32_t mask, int func, uint32_t nparts,
                    uintptr_t stream);
size_t sort_workspace_bytes(uint32_t n);
int sort_pairs_u64(uintptr_t keys, uintptr_t vals, uintptr_t tmp_keys,
                   uintptr_t tmp_vals, uint32_t n, int start_bit, int end_bit,
                   uintptr_t ws, uintptr_t stream);
void probe_scatter_write(uintptr_t out, uint64_t region_elems,
                         uint32_t nregions, uint32_t burst_elems,
                         uint32_t bursts_per_block, uint32_t grid,
                         uintptr_t stream);
void join_count(uintptr_t a_keys, uint32_t na, uintptr_t b_keys,
                uint32_t nb_, uintptr_t counts, uintptr_t lo_idx,
                uintptr_t stream);
void join_emit(uintptr_t a_keys, uintptr_t a_vals, uint32_t na,
               uintptr_t b_vals, uintptr_t counts, uintptr_t lo_idx,
               uintptr_t offsets, uintptr_t out_key, uintptr_t out_a,
               uintptr_t out_b, uintptr_t stream);
size_t onesweep_workspace_bytes(uint32_t n, int passes);
void set_aos_tile(int t);
void set_pass_stage(int s);
void set_timing_buf(uintptr_t p);
void set_sort_mode(int m);
void set_lookback_mode(int m);
void set_split_exchange(int m);
void set_lean_pass(int m);
int onesweep_sort_aos7_u64(uintptr_t pairs, uintptr_t tmp_pairs, uint32_t n,
                           int start_bit, int end_bit, uintptr_t ws,
                           uintptr_t stream);
int onesweep_sort_aos_u64(uintptr_t pairs, uintptr_t tmp_pairs, uint32_t n,
                          int start_bit, int end_bit, uintptr_t ws,
                          uintptr_t stream);
int onesweep_sort_aos_word_u64(uintptr_t pairs, uintptr_t tmp_pairs,
                               uint32_t n, int start_bit, int end_bit,
                               uintptr_t ws, uintptr_t stream,
                               int sort_word);
int onesweep_sort_aos_fused_u64(uintptr_t pairs, uintptr_t tmp_pairs,
                                uint32_t n, int start_bit, int end_bit,
                                uintptr_t ws, uintptr_t stream,
                                int sort_word, uintptr_t recs,
                                uintptr_t out, uint32_t rec_bytes);
int onesweep_sort_pairs_u64(uintptr_t keys, uintptr_t vals,
                            uintptr_t tmp_keys, uintptr_t tmp_vals,
                            uint32_t n, int start_bit, int end_bit,
                            uintptr_t ws, uintptr_t stream);

}  // namespace hipshuffle
