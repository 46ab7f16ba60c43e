#include "blackbird/client/shuffle.h"

#include <condition_variable>
#include <cstring>
#include <mutex>

#include <hip/hip_runtime_api.h>

#include "blackbird/common/log.h"
#include "blackbird/common/serde.h"
#include "blackbird/common/trace.h"
#include "blackbird/transport/rccl_engine.h"

namespace blackbird {

namespace {
constexpr uint64_t kErrTotal = ~0ull;  // "my resolution failed" sentinel

Error hip_err(hipError_t e, const char* what) {
  return Error{ErrorCode::HIP_ERROR,
               std::string(what) + ": " + hipGetErrorString(e)};
}
#define BB_HIP(expr)                                  \
  do {                                                \
    hipError_t _e = (expr);                           \
    if (_e != hipSuccess) return hip_err(_e, #expr);  \
  } while (0)

struct TransportBuf {  // RAII transport-memory allocation
  Copier& cp;
  void* p = nullptr;
  TransportBuf(Copier& c, uint64_t n) : cp(c) {
    if (n) {
      auto r = c.alloc(n);
      if (r.ok()) p = r.value();
    }
  }
  ~TransportBuf() {
    if (p) cp.free(p);
  }
};
}  // namespace

Result<void> batch_shuffle(Exchanger& ex, Copier& cp,
                           const ShuffleResolver& resolve,
                           const std::vector<ShuffleWant>& want) {
  BB_TRACE_SCOPE("bb::batch_shuffle");
  const int n = ex.nranks();
  const int me = ex.rank();
  if (static_cast<int>(want.size()) != n)
    return Error{ErrorCode::INVALID_ARGUMENT, "want list needs nranks slots"};

  // ---- phase 1: encode + exchange want lists ----
  // blob per peer: u32 count, then (str key, u64 size) pairs
  std::vector<std::string> blobs(n);
  std::vector<uint64_t> expect_recv(n, 0);  // bytes I expect from peer p
  for (int p = 0; p < n; ++p) {
    const auto& w = want[p];
    if (w.keys.size() != w.sizes.size())
      return Error{ErrorCode::INVALID_ARGUMENT, "keys/sizes length mismatch"};
    serde::Enc e;
    e.num<uint32_t>(static_cast<uint32_t>(w.keys.size()));
    for (size_t i = 0; i < w.keys.size(); ++i) {
      e.str(w.keys[i]);
      e.num<uint64_t>(w.sizes[i]);
      expect_recv[p] += w.sizes[i];
    }
    if (expect_recv[p] > 0 && w.recv_base == nullptr)
      return Error{ErrorCode::INVALID_ARGUMENT, "missing recv buffer"};
    blobs[p] = std::move(e.buf);
  }

  // lengths (u64 per peer), in transport memory
  std::vector<uint64_t> send_lens(n), recv_lens(n, 0);
  for (int p = 0; p < n; ++p) send_lens[p] = blobs[p].size();
  TransportBuf slen(cp, n * 8), rlen(cp, n * 8);
  if (!slen.p || !rlen.p)
    return Error{ErrorCode::INTERNAL_ERROR, "shuffle staging alloc"};
  BB_RETURN_IF_ERROR(cp.to_transport(slen.p, send_lens.data(), n * 8));
  {
    std::vector<const void*> sp(n);
    std::vector<void*> rp(n);
    std::vector<uint64_t> sb(n, 8), rb(n, 8);
    for (int p = 0; p < n; ++p) {
      sp[p] = static_cast<uint8_t*>(slen.p) + p * 8;
      rp[p] = static_cast<uint8_t*>(rlen.p) + p * 8;
    }
    BB_RETURN_IF_ERROR(ex.alltoallv(sp, sb, rp, rb));
  }
  BB_RETURN_IF_ERROR(cp.from_transport(recv_lens.data(), rlen.p, n * 8));

  // blobs
  uint64_t send_blob_total = 0, recv_blob_total = 0;
  for (int p = 0; p < n; ++p) {
    send_blob_total += send_lens[p];
    recv_blob_total += recv_lens[p];
  }
  TransportBuf sblob(cp, send_blob_total), rblob(cp, recv_blob_total);
  if ((send_blob_total && !sblob.p) || (recv_blob_total && !rblob.p))
    return Error{ErrorCode::INTERNAL_ERROR, "shuffle staging alloc"};
  {
    uint64_t off = 0;
    for (int p = 0; p < n; ++p) {
      if (send_lens[p])
        BB_RETURN_IF_ERROR(cp.to_transport(
            static_cast<uint8_t*>(sblob.p) + off, blobs[p].data(),
            send_lens[p]));
      off += send_lens[p];
    }
    std::vector<const void*> sp(n);
    std::vector<void*> rp(n);
    uint64_t so = 0, ro = 0;
    for (int p = 0; p < n; ++p) {
      sp[p] = static_cast<uint8_t*>(sblob.p) + so;
      rp[p] = static_cast<uint8_t*>(rblob.p) + ro;
      so += send_lens[p];
      ro += recv_lens[p];
    }
    BB_RETURN_IF_ERROR(ex.alltoallv(sp, send_lens, rp, recv_lens));
  }

  // ---- phase 2: decode serve lists, resolve local objects, gather ----
  std::string rhost(recv_blob_total, '\0');
  if (recv_blob_total)
    BB_RETURN_IF_ERROR(cp.from_transport(rhost.data(), rblob.p,
                                         recv_blob_total));
  bool my_ok = true;
  std::vector<uint64_t> serve_total(n, 0);
  std::vector<gpu::CopyDesc> gather;
  uint64_t stage_needed = 0;
  {
    // first pass: sizes (to lay out the send staging buffer)
    uint64_t off = 0;
    for (int p = 0; p < n && my_ok; ++p) {
      serde::Dec d(rhost.data() + off, recv_lens[p]);
      off += recv_lens[p];
      uint32_t cnt = d.num<uint32_t>();
      for (uint32_t i = 0; i < cnt && d.ok(); ++i) {
        d.str();
        serve_total[p] += d.num<uint64_t>();
      }
      if (!d.ok()) my_ok = false;
      stage_needed += serve_total[p];
    }
  }
  TransportBuf stage(cp, my_ok ? stage_needed : 0);
  if (my_ok && stage_needed && !stage.p) my_ok = false;
  if (my_ok) {
    uint64_t off = 0, soff = 0;
    for (int p = 0; p < n && my_ok; ++p) {
      serde::Dec d(rhost.data() + off, recv_lens[p]);
      off += recv_lens[p];
      uint32_t cnt = d.num<uint32_t>();
      for (uint32_t i = 0; i < cnt && d.ok(); ++i) {
        std::string key = d.str();
        uint64_t sz = d.num<uint64_t>();
        const void* src = d.ok() ? resolve(key, sz) : nullptr;
        if (!src) {
          my_ok = false;
          break;
        }
        gather.push_back({src, static_cast<uint8_t*>(stage.p) + soff, sz});
        soff += sz;
      }
      if (!d.ok()) my_ok = false;
    }
  }
  if (my_ok && !gather.empty()) {
    auto r = cp.copy_batch(gather.data(), static_cast<uint32_t>(gather.size()));
    if (!r.ok()) my_ok = false;
  }

  // ---- phase 3: totals handshake (consistent abort instead of a hang) ----
  // a failed rank sends kErrTotal to EVERY peer; any receiver of kErrTotal
  // (or of a total that disagrees with its own want list) aborts — and since
  // the failure was broadcast, every rank aborts together
  std::vector<uint64_t> tot_out(n), tot_in(n, 0);
  for (int p = 0; p < n; ++p) tot_out[p] = my_ok ? serve_total[p] : kErrTotal;
  TransportBuf stot(cp, n * 8), rtot(cp, n * 8);
  if (!stot.p || !rtot.p)
    return Error{ErrorCode::INTERNAL_ERROR, "shuffle staging alloc"};
  BB_RETURN_IF_ERROR(cp.to_transport(stot.p, tot_out.data(), n * 8));
  {
    std::vector<const void*> sp(n);
    std::vector<void*> rp(n);
    std::vector<uint64_t> sb(n, 8), rb(n, 8);
    for (int p = 0; p < n; ++p) {
      sp[p] = static_cast<uint8_t*>(stot.p) + p * 8;
      rp[p] = static_cast<uint8_t*>(rtot.p) + p * 8;
    }
    BB_RETURN_IF_ERROR(ex.alltoallv(sp, sb, rp, rb));
  }
  BB_RETURN_IF_ERROR(cp.from_transport(tot_in.data(), rtot.p, n * 8));
  if (!my_ok)
    return Error{ErrorCode::OBJECT_NOT_FOUND,
                 "shuffle: could not serve a requested object"};
  for (int p = 0; p < n; ++p)
    if (tot_in[p] == kErrTotal || tot_in[p] != expect_recv[p])
      return Error{ErrorCode::SIZE_MISMATCH,
                   "shuffle aborted: peer " + std::to_string(p) +
                       (tot_in[p] == kErrTotal ? " failed resolution"
                                               : " size disagreement")};

  // ---- phase 4: the data all-to-all ----
  {
    std::vector<const void*> sp(n);
    std::vector<void*> rp(n);
    uint64_t soff = 0;
    for (int p = 0; p < n; ++p) {
      sp[p] = static_cast<uint8_t*>(stage.p) + soff;
      soff += serve_total[p];
      rp[p] = want[p].recv_base;
    }
    BB_RETURN_IF_ERROR(ex.alltoallv(sp, serve_total, rp, expect_recv));
  }
  return {};
}

// --------------------------------------------------------------- RCCL/GPU

int RcclExchanger::rank() const { return e_.rank(); }
int RcclExchanger::nranks() const { return e_.nranks(); }

Result<void> RcclExchanger::alltoallv(const std::vector<const void*>& sp,
                                      const std::vector<uint64_t>& sb,
                                      const std::vector<void*>& rp,
                                      const std::vector<uint64_t>& rb) {
  BB_RETURN_IF_ERROR(e_.alltoallv(sp, sb, rp, rb, stream_));
  BB_HIP(hipStreamSynchronize(stream_));
  return {};
}

Result<void*> GpuCopier::alloc(uint64_t nbytes) {
  BB_HIP(hipSetDevice(device_));
  void* p = nullptr;
  BB_HIP(hipMalloc(&p, nbytes));
  return p;
}

void GpuCopier::free(void* p) { (void)hipFree(p); }

Result<void> GpuCopier::copy_batch(const gpu::CopyDesc* descs, uint32_t n) {
  BB_RETURN_IF_ERROR(gpu::batched_copy(descs, n, stream_));
  BB_HIP(hipStreamSynchronize(stream_));
  return {};
}

Result<void> GpuCopier::to_transport(void* dst, const void* src, uint64_t n) {
  BB_RETURN_IF_ERROR(gpu::copy_sync(dst, src, n, hipMemcpyHostToDevice));
  return {};
}

Result<void> GpuCopier::from_transport(void* dst, const void* src, uint64_t n) {
  BB_RETURN_IF_ERROR(gpu::copy_sync(dst, src, n, hipMemcpyDeviceToHost));
  return {};
}

// ---------------------------------------------------------- host loopback

struct LoopbackGroup::State {
  std::mutex mu;
  std::condition_variable cv;
  int nranks;
  int arrived = 0;
  uint64_t gen = 0;
  bool error = false;
  struct Post {
    const std::vector<const void*>* sp = nullptr;
    const std::vector<uint64_t>* sb = nullptr;
  };
  std::vector<Post> posts;

  explicit State(int n) : nranks(n), posts(n) {}

  // returns false if the group flagged an error this round
  bool barrier() {
    std::unique_lock<std::mutex> lk(mu);
    uint64_t g = gen;
    if (++arrived == nranks) {
      arrived = 0;
      ++gen;
      cv.notify_all();
    } else {
      cv.wait(lk, [&] { return gen != g; });
    }
    return !error;
  }
};

class LoopbackGroup::Rank : public Exchanger {
 public:
  Rank(std::shared_ptr<State> st, int rank) : st_(std::move(st)), rank_(rank) {}
  int rank() const override { return rank_; }
  int nranks() const override { return st_->nranks; }

  Result<void> alltoallv(const std::vector<const void*>& sp,
                         const std::vector<uint64_t>& sb,
                         const std::vector<void*>& rp,
                         const std::vector<uint64_t>& rb) override {
    {
      std::lock_guard<std::mutex> g(st_->mu);
      st_->posts[rank_] = {&sp, &sb};
    }
    st_->barrier();  // everyone posted
    bool ok = true;
    for (int p = 0; p < st_->nranks; ++p) {
      const auto& post = st_->posts[p];
      if ((*post.sb)[rank_] != rb[p]) {
        ok = false;
        break;
      }
      if (rb[p] > 0) std::memcpy(rp[p], (*post.sp)[rank_], rb[p]);
    }
    if (!ok) {
      std::lock_guard<std::mutex> g(st_->mu);
      st_->error = true;
    }
    // pulls done; posts may be released. A size mismatch poisons the group
    // for its remaining lifetime (tests build a fresh group per exchange).
    bool group_ok = st_->barrier();
    if (!ok || !group_ok)
      return Error{ErrorCode::SIZE_MISMATCH, "loopback alltoallv mismatch"};
    return {};
  }

 private:
  std::shared_ptr<State> st_;
  int rank_;
};

LoopbackGroup::LoopbackGroup(int nranks)
    : nranks_(nranks), st_(std::make_shared<State>(nranks)) {}

LoopbackGroup::~LoopbackGroup() = default;

std::unique_ptr<Exchanger> LoopbackGroup::exchanger(int rank) {
  return std::make_unique<Rank>(st_, rank);
}

Result<void*> HostCopier::alloc(uint64_t nbytes) {
  void* p = ::malloc(nbytes);
  if (!p) return Error{ErrorCode::INTERNAL_ERROR, "malloc"};
  return p;
}

void HostCopier::free(void* p) { ::free(p); }

Result<void> HostCopier::copy_batch(const gpu::CopyDesc* descs, uint32_t n) {
  for (uint32_t i = 0; i < n; ++i)
    std::memcpy(descs[i].dst, descs[i].src, descs[i].nbytes);
  return {};
}

Result<void> HostCopier::to_transport(void* dst, const void* src, uint64_t n) {
  std::memcpy(dst, src, n);
  return {};
}

Result<void> HostCopier::from_transport(void* dst, const void* src,
                                        uint64_t n) {
  std::memcpy(dst, src, n);
  return {};
}

}  // namespace blackbird
