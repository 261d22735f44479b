// HBM slab backend + xGMI peer-copy engine.
//
// MI355X-native replacement for the reference's verbs primitives
// (SURVEY.md §2.3 mapping table):
//   ibv_reg_mr                -> hipMalloc slab + hipIpcGetMemHandle
//   rdma_cm connect           -> hipIpcOpenMemHandle + enable peer access
//   IBV_WR_RDMA_READ scatter  -> hipMemcpyAsync D2D over xGMI on a per-peer
//                                stream; the "signaled last WR" is a
//                                hipEvent recorded after the batch
//   CQ poller                 -> event queries from the Python fetch pool
//
// Allocation POLICY (buddy classes, trim) stays in tested Python
// (block_pool.py); this layer only provides slabs and copies.

#include "common.h"

#include <sys/socket.h>

#include <atomic>
#include <cstring>
#include <mutex>
#include <unordered_map>
#include <vector>

namespace hipshuffle {

// ---------------------------------------------------------------------------
// slabs

struct Slab {
  void* ptr = nullptr;
  size_t size = 0;
  hipIpcMemHandle_t handle{};
};

class SlabPool {
 public:
  // Returns (slab_id). Handle bytes retrievable via handle_bytes().
  int alloc(size_t bytes) {
    Slab s;
    HIP_CHECK(hipMalloc(&s.ptr, bytes));
    s.size = bytes;
    HIP_CHECK(hipIpcGetMemHandle(&s.handle, s.ptr));
    std::lock_guard<std::mutex> g(mu_);
    int id = next_id_++;
    slabs_[id] = s;
    return id;
  }

  void free(int id) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = slabs_.find(id);
    if (it == slabs_.end()) throw std::runtime_error("bad slab id");
    HIP_CHECK(hipFree(it->second.ptr));
    slabs_.erase(it);
  }

  // By VALUE: a reference into the map would dangle if a concurrent
  // free() erases/rehashes after the lock drops.
  Slab get(int id) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = slabs_.find(id);
    if (it == slabs_.end()) throw std::runtime_error("bad slab id");
    return it->second;
  }

  uintptr_t base(int id) { return reinterpret_cast<uintptr_t>(get(id).ptr); }

  std::string handle_bytes(int id) {
    Slab s = get(id);
    return std::string(reinterpret_cast<const char*>(&s.handle),
                       sizeof(hipIpcMemHandle_t));
  }

 private:
  std::mutex mu_;
  std::unordered_map<int, Slab> slabs_;
  int next_id_ = 0;
};

SlabPool& slab_pool() {
  static SlabPool p;
  return p;
}

// ---------------------------------------------------------------------------
// imported (peer) slabs — opened IPC handles, cached by the Python side

uintptr_t ipc_open(const std::string& handle_bytes) {
  if (handle_bytes.size() != sizeof(hipIpcMemHandle_t))
    throw std::runtime_error("bad ipc handle size");
  hipIpcMemHandle_t h;
  std::memcpy(&h, handle_bytes.data(), sizeof(h));
  void* ptr = nullptr;
  HIP_CHECK(hipIpcOpenMemHandle(&ptr, h, hipIpcMemLazyEnablePeerAccess));
  return reinterpret_cast<uintptr_t>(ptr);
}

void ipc_close(uintptr_t ptr) {
  HIP_CHECK(hipIpcCloseMemHandle(reinterpret_cast<void*>(ptr)));
}

void enable_peer_access(int peer_device) {
  hipError_t e = hipDeviceEnablePeerAccess(peer_device, 0);
  if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) HIP_CHECK(e);
}

// ---------------------------------------------------------------------------
// per-peer copy engine: streams + recycled completion events
//
// The reference posts a scatter list of READ WRs and signals only the last
// (RdmaChannel.java:484-517). Here: N hipMemcpyAsync on one of the peer's
// streams, then one recorded event = the signaled WR. Completion is polled
// (hipEventQuery) from the Python completion thread — the CQ-poller analog
// (RdmaThread.java:45-58).
//
// r02 changes (VERDICT r01 "event churn", "streams by peer GPU"):
//   * events come from a free list and are RECYCLED, never created or
//     destroyed per batch — the analog of the reference caching its
//     serialized verbs objects (RdmaChannel.java:43-44,370-420).
//   * `peer` is the peer's GPU ordinal (the xGMI link selector), not an
//     executor id; four streams per peer rotate so several SDMA engines
//     can drive one link/local-HBM concurrently while different peers'
//     copies never serialize behind each other.

struct PeerEngine {
  static constexpr int kStreams = 4;
  hipStream_t streams[kStreams] = {};
  unsigned next = 0;
  std::mutex mu;
};

class CopyEngine {
 public:
  static constexpr int kMaxPeers = 64;

  CopyEngine() : peers_(kMaxPeers) {}

  hipStream_t stream_for(int peer) {
    PeerEngine& p = peers_.at(peer);
    std::lock_guard<std::mutex> g(p.mu);
    unsigned i = p.next++ % PeerEngine::kStreams;
    if (!p.streams[i])
      HIP_CHECK(hipStreamCreateWithFlags(&p.streams[i], hipStreamNonBlocking));
    return p.streams[i];
  }

  // Issue a batch of copies; returns an event id to poll.
  // hipMemcpyDefault: the pointers select the direction, so the same
  // engine serves xGMI D2D reads AND pinned-host staging (TCP lane).
  uint64_t read_batch(int peer, const std::vector<uintptr_t>& dsts,
                      const std::vector<uintptr_t>& srcs,
                      const std::vector<size_t>& sizes) {
    hipStream_t s = stream_for(peer);
    for (size_t i = 0; i < dsts.size(); ++i) {
      HIP_CHECK(hipMemcpyAsync(reinterpret_cast<void*>(dsts[i]),
                               reinterpret_cast<void*>(srcs[i]), sizes[i],
                               hipMemcpyDefault, s));
    }
    hipEvent_t ev = acquire_event();
    HIP_CHECK(hipEventRecord(ev, s));
    uint64_t id = next_event_.fetch_add(1);
    std::lock_guard<std::mutex> g(ev_mu_);
    events_[id] = ev;
    return id;
  }

  // true = complete (event recycled), false = still in flight
  bool poll(uint64_t id) {
    hipEvent_t ev;
    {
      std::lock_guard<std::mutex> g(ev_mu_);
      auto it = events_.find(id);
      if (it == events_.end()) return true;  // already reaped
      ev = it->second;
    }
    hipError_t e = hipEventQuery(ev);
    if (e == hipErrorNotReady) return false;
    HIP_CHECK(e);
    release_event(id, ev);
    return true;
  }

  void wait(uint64_t id) {
    hipEvent_t ev;
    {
      std::lock_guard<std::mutex> g(ev_mu_);
      auto it = events_.find(id);
      if (it == events_.end()) return;
      ev = it->second;
    }
    HIP_CHECK(hipEventSynchronize(ev));
    release_event(id, ev);
  }

 private:
  hipEvent_t acquire_event() {
    {
      std::lock_guard<std::mutex> g(ev_mu_);
      if (!free_events_.empty()) {
        hipEvent_t ev = free_events_.back();
        free_events_.pop_back();
        return ev;
      }
    }
    hipEvent_t ev;
    HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    return ev;
  }

  void release_event(uint64_t id, hipEvent_t ev) {
    std::lock_guard<std::mutex> g(ev_mu_);
    // erase() guards concurrent poll()/wait() reaps of the same id: only
    // the first releaser recycles the event; a loser that then queries a
    // re-acquired event can at worst see NotReady once more, never a
    // false completion (its id is gone from the map -> "already reaped").
    if (events_.erase(id)) free_events_.push_back(ev);
  }

  std::vector<PeerEngine> peers_;
  std::atomic<uint64_t> next_event_{1};
  std::mutex ev_mu_;
  std::unordered_map<uint64_t, hipEvent_t> events_;
  std::vector<hipEvent_t> free_events_;
};

CopyEngine& copy_engine() {
  static CopyEngine e;
  return e;
}

// flat C++ surface (hipshuffle.h)
int slab_alloc_id(size_t bytes) { return slab_pool().alloc(bytes); }
void slab_free_id(int id) { slab_pool().free(id); }
uintptr_t slab_base_id(int id) { return slab_pool().base(id); }
std::string slab_handle_id(int id) { return slab_pool().handle_bytes(id); }
uint64_t read_batch_ids(int peer, const std::vector<uintptr_t>& dsts,
                        const std::vector<uintptr_t>& srcs,
                        const std::vector<size_t>& sizes) {
  return copy_engine().read_batch(peer, dsts, srcs, sizes);
}
bool poll_event(uint64_t id) { return copy_engine().poll(id); }
void wait_event(uint64_t id) { return copy_engine().wait(id); }

// pinned host memory for the TCP lane's double-buffered D2H staging
uintptr_t host_alloc_pinned(size_t n) {
  void* p = nullptr;
  HIP_CHECK(hipHostMalloc(&p, n, hipHostMallocDefault));
  return reinterpret_cast<uintptr_t>(p);
}

void host_free_pinned(uintptr_t p) {
  HIP_CHECK(hipHostFree(reinterpret_cast<void*>(p)));
}

// ---------------------------------------------------------------------------
// GIL-free socket helpers for the cross-host TCP lane: the chunk-framed
// receive loop and raw sends run entirely in C (plain POSIX fds), so the
// per-process lane bandwidth is not bound by Python byte handling.

int64_t tcp_recv_chunks(int fd, uintptr_t out, uint64_t total) {
  // reads [raw_len u32][wire_len u32][payload] frames until `total` raw
  // bytes landed; only UNCOMPRESSED frames (wire==raw) are handled here
  // — the deflate path stays in Python. Returns bytes received, or
  // -1 on EOF/framing error, -2 if a compressed frame appears (caller
  // must not use this helper when the codec is on).
  uint8_t* dst = reinterpret_cast<uint8_t*>(out);
  uint64_t off = 0;
  while (off < total) {
    uint32_t hdr[2];
    size_t got = 0;
    while (got < sizeof(hdr)) {
      ssize_t r = ::recv(fd, reinterpret_cast<char*>(hdr) + got,
                         sizeof(hdr) - got, 0);
      if (r <= 0) return -1;
      got += (size_t)r;
    }
    uint32_t raw_len = hdr[0], wire_len = hdr[1];
    if (wire_len != raw_len) return -2;
    if (off + raw_len > total) return -1;
    uint64_t done = 0;
    while (done < raw_len) {
      ssize_t r = ::recv(fd, dst + off + done, raw_len - done, 0);
      if (r <= 0) return -1;
      done += (uint64_t)r;
    }
    off += raw_len;
  }
  return (int64_t)off;
}

int64_t tcp_send_all(int fd, uintptr_t buf, uint64_t n) {
  const uint8_t* p = reinterpret_cast<const uint8_t*>(buf);
  uint64_t off = 0;
  while (off < n) {
    ssize_t r = ::send(fd, p + off, n - off, MSG_NOSIGNAL);
    if (r <= 0) return -1;
    off += (uint64_t)r;
  }
  return (int64_t)off;
}

// host<->device staging helpers (bytes path / spill)
void memcpy_h2d(uintptr_t dst, uintptr_t src, size_t n) {
  HIP_CHECK(hipMemcpy(reinterpret_cast<void*>(dst),
                      reinterpret_cast<void*>(src), n, hipMemcpyHostToDevice));
}

void memcpy_d2h(uintptr_t dst, uintptr_t src, size_t n) {
  HIP_CHECK(hipMemcpy(reinterpret_cast<void*>(dst),
                      reinterpret_cast<void*>(src), n, hipMemcpyDeviceToHost));
}

}  // namespace hipshuffle
