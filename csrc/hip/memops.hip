// Batched scatter/gather copy + test-pattern kernels for the HBM tier
// (gfx950). One launch moves a whole batch of objects between slab locations
// and staging/ring buffers — the fused multi-object path behind
// batch_put/batch_get (reference analogue: the per-shard ucp_put_nbx loop in
// blackbird_client.cpp:252-267, replaced here by one kernel over all shards).
//
// CDNA4 notes: 16 B/lane dwordx4 vector moves, grid-stride over 4 KiB chunks
// so a batch of any shape fills 256 CUs; descriptors are binary-searched per
// chunk (L2-resident, lane-uniform per wave).
#include <hip/hip_runtime.h>
#include <cstdlib>
#include <cstring>

#include <atomic>
#include <memory>
#include <mutex>
#include <vector>

#include "blackbird/gpu/digest_spec.h"
#include "digest_device.hip.h"
#include "blackbird/gpu/gpu_kernels.h"
#include "hip_common.h"

namespace blackbird::gpu {

namespace {

constexpr int kBlock = 256;
constexpr uint64_t kChunk = 4096;  // copy work quantum

struct Seg {
  const uint8_t* src;
  uint8_t* dst;
  uint64_t nbytes;
};

using u8 = uint8_t;
using ulong1 = unsigned long long;
using v16 = __attribute__((ext_vector_type(4))) unsigned int;

__global__ void __launch_bounds__(kBlock)
batched_copy_kernel(const Seg* __restrict__ segs,
                    const uint64_t* __restrict__ chunk_prefix, uint32_t nsegs,
                    uint64_t total_chunks) {
  const uint64_t gid0 = static_cast<uint64_t>(blockIdx.x) * kBlock + threadIdx.x;
  const uint64_t wave_id = gid0 >> 6;          // one chunk per wave
  const int lane = threadIdx.x & 63;
  const uint64_t wave_stride =
      (static_cast<uint64_t>(gridDim.x) * kBlock) >> 6;

  for (uint64_t c = wave_id; c < total_chunks; c += wave_stride) {
    // find segment: largest i with chunk_prefix[i] <= c
    uint32_t lo = 0, hi = nsegs - 1;
    while (lo < hi) {
      uint32_t mid = (lo + hi + 1) >> 1;
      if (chunk_prefix[mid] <= c) lo = mid;
      else hi = mid - 1;
    }
    const Seg s = segs[lo];
    const uint64_t off = (c - chunk_prefix[lo]) * kChunk;
    const uint64_t len = min(kChunk, s.nbytes - off);
    const uint8_t* src = s.src + off;
    uint8_t* dst = s.dst + off;

    const bool aligned = ((reinterpret_cast<uintptr_t>(src) |
                           reinterpret_cast<uintptr_t>(dst)) & 15) == 0;
    if (aligned) {
      // 16 B per lane: 64 lanes × 16 = 1024 B per iteration; nontemporal —
      // streamed bytes are never re-read from cache by this kernel
      uint64_t nvec = len >> 4;
      for (uint64_t i = lane; i < nvec; i += 64) {
        v16 v = __builtin_nontemporal_load(reinterpret_cast<const v16*>(src) + i);
        __builtin_nontemporal_store(v, reinterpret_cast<v16*>(dst) + i);
      }
      // tail bytes
      for (uint64_t i = (nvec << 4) + lane; i < len; i += 64) dst[i] = src[i];
    } else {
      for (uint64_t i = lane; i < len; i += 64) dst[i] = src[i];
    }
  }
}

__global__ void __launch_bounds__(kBlock)
fill_kernel(ulong1* __restrict__ out, uint64_t nwords, uint64_t seed) {
  const uint64_t stride = static_cast<uint64_t>(gridDim.x) * kBlock;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * kBlock + threadIdx.x;
       i < nwords; i += stride)
    out[i] = digest::splitmix64(seed ^ i);
}

__global__ void __launch_bounds__(kBlock)
verify_kernel(const ulong1* __restrict__ in, uint64_t nwords, uint64_t seed,
              ulong1* __restrict__ bad) {
  uint64_t local = 0;
  const uint64_t stride = static_cast<uint64_t>(gridDim.x) * kBlock;
  for (uint64_t i = static_cast<uint64_t>(blockIdx.x) * kBlock + threadIdx.x;
       i < nwords; i += stride)
    if (in[i] != digest::splitmix64(seed ^ i)) ++local;
  if (local) atomicAdd(bad, local);
}

int grid_for(uint64_t work_items) {
  uint64_t blocks = (work_items + kBlock - 1) / kBlock;
  if (blocks < 1) blocks = 1;
  if (blocks > 4096) blocks = 4096;
  return static_cast<int>(blocks);
}

}  // namespace

namespace {
// Persistent descriptor staging. batched_copy() returns BEFORE the stream
// executes, so both the host image AND the device descriptor buffer must
// outlive the call: a per-thread {pinned, device} pair is grown once and
// reused; an event guards against overwriting an upload that a previous
// launch still reads.
struct DescStaging {
  void* pinned = nullptr;
  void* device = nullptr;
  size_t cap = 0;
  hipEvent_t ev = nullptr;

  Result<void> acquire(size_t bytes) {
    if (ev) BB_HIP_TRY(hipEventSynchronize(ev));  // prior launch done
    else BB_HIP_TRY(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    if (bytes > cap) {
      if (pinned) BB_HIP_TRY(hipHostFree(pinned));
      if (device) BB_HIP_TRY(hipFree(device));
      cap = std::max<size_t>(bytes * 2, 1 << 16);
      BB_HIP_TRY(hipHostMalloc(&pinned, cap, hipHostMallocDefault));
      BB_HIP_TRY(hipMalloc(&device, cap));
    }
    return {};
  }
};
thread_local DescStaging g_copy_stage;
}  // namespace

Result<void> batched_copy(const CopyDesc* descs, uint32_t n, hipStream_t stream) {
  if (n == 0) return {};
  const size_t segs_bytes = n * sizeof(Seg);
  const size_t prefix_bytes = n * sizeof(uint64_t);
  BB_RETURN_IF_ERROR(g_copy_stage.acquire(segs_bytes + prefix_bytes));
  Seg* h_segs = static_cast<Seg*>(g_copy_stage.pinned);
  uint64_t* h_prefix = reinterpret_cast<uint64_t*>(
      static_cast<uint8_t*>(g_copy_stage.pinned) + segs_bytes);

  uint64_t total = 0;
  for (uint32_t i = 0; i < n; ++i) {
    h_segs[i] = {static_cast<const uint8_t*>(descs[i].src),
                 static_cast<uint8_t*>(descs[i].dst), descs[i].nbytes};
    h_prefix[i] = total;
    total += (descs[i].nbytes + kChunk - 1) / kChunk;
  }
  if (total == 0) return {};

  Seg* d_segs = static_cast<Seg*>(g_copy_stage.device);
  uint64_t* d_prefix = reinterpret_cast<uint64_t*>(
      static_cast<uint8_t*>(g_copy_stage.device) + segs_bytes);
  BB_HIP_TRY(hipMemcpyAsync(d_segs, h_segs, segs_bytes + prefix_bytes,
                            hipMemcpyHostToDevice, stream));
  // one wave per chunk → want total waves ≈ total chunks
  const int blocks = grid_for(total * 64);
  batched_copy_kernel<<<blocks, kBlock, 0, stream>>>(d_segs, d_prefix, n, total);
  if (hipError_t _le = hipGetLastError(); _le != hipSuccess)
    return hip_error(_le, "batched_copy_kernel launch");
  BB_HIP_TRY(hipEventRecord(g_copy_stage.ev, stream));
  return {};
}

Result<void> fill_pattern(void* dev_ptr, uint64_t nbytes, uint64_t seed,
                          hipStream_t stream) {
  const uint64_t nwords = nbytes / 8;
  if (nwords > 0) {
    fill_kernel<<<grid_for(nwords / 4), kBlock, 0, stream>>>(
        static_cast<ulong1*>(dev_ptr), nwords, seed);
    if (hipError_t _le = hipGetLastError(); _le != hipSuccess)
      return hip_error(_le, "fill_kernel launch");
  }
  return {};
}

Result<uint64_t> verify_pattern(const void* dev_ptr, uint64_t nbytes, uint64_t seed,
                                hipStream_t stream) {
  const uint64_t nwords = nbytes / 8;
  ulong1* d_bad = nullptr;
  BB_HIP_TRY(hipMalloc(reinterpret_cast<void**>(&d_bad), 8));
  BB_HIP_TRY(hipMemsetAsync(d_bad, 0, 8, stream));
  if (nwords > 0) {
    verify_kernel<<<grid_for(nwords / 4), kBlock, 0, stream>>>(
        static_cast<const ulong1*>(dev_ptr), nwords, seed, d_bad);
    if (hipError_t _le = hipGetLastError(); _le != hipSuccess)
      return hip_error(_le, "verify_kernel launch");
  }
  uint64_t bad = 0;
  BB_HIP_TRY(hipMemcpyAsync(&bad, d_bad, 8, hipMemcpyDeviceToHost, stream));
  BB_HIP_TRY(hipStreamSynchronize(stream));
  BB_HIP_TRY(hipFree(d_bad));
  return bad;
}

}  // namespace blackbird::gpu

namespace blackbird::gpu {

namespace {
// Shared pool of pre-created non-blocking transfer streams. The library
// never touches the legacy (null) stream — a null-stream op in ANY thread
// fails while a hipGraph capture is open — but streams must NOT be
// per-thread: the staged fan-out paths run on short-lived std::async
// threads, and creating/destroying a stream per thread costs milliseconds
// on ROCm (measured 9× on the NVMe get leg). Round-robin over a fixed pool
// instead; a per-stream mutex keeps each async+sync pair atomic.
constexpr int kXferStreams = 8;
struct XferPool {
  hipStream_t s[kXferStreams] = {};
  std::mutex mu[kXferStreams];
  std::atomic<uint32_t> next{0};
  std::once_flag once;
  hipError_t init_rc = hipSuccess;

  hipError_t ensure() {
    std::call_once(once, [this] {
      for (int i = 0; i < kXferStreams; ++i) {
        init_rc = hipStreamCreateWithFlags(&s[i], hipStreamNonBlocking);
        if (init_rc != hipSuccess) return;
      }
    });
    return init_rc;
  }
};
XferPool g_xfer;
}  // namespace

Result<void> copy_sync(void* dst, const void* src, uint64_t nbytes, int kind) {
  if (nbytes == 0) return {};
  static const bool legacy = std::getenv("BB_COPY_LEGACY") != nullptr;
  if (legacy) {  // A/B diagnostics only: null-stream copies break captures
    BB_HIP_TRY(hipMemcpy(dst, src, nbytes, static_cast<hipMemcpyKind>(kind)));
    return {};
  }
  BB_HIP_TRY(g_xfer.ensure());
  const uint32_t i = g_xfer.next.fetch_add(1) % kXferStreams;
  std::lock_guard<std::mutex> g(g_xfer.mu[i]);
  BB_HIP_TRY(hipMemcpyAsync(dst, src, nbytes,
                            static_cast<hipMemcpyKind>(kind), g_xfer.s[i]));
  BB_HIP_TRY(hipStreamSynchronize(g_xfer.s[i]));
  return {};
}

Result<uint64_t> device_malloc(uint64_t nbytes, int device) {
  BB_HIP_TRY(hipSetDevice(device));
  void* p = nullptr;
  BB_HIP_TRY(hipMalloc(&p, nbytes));
  return reinterpret_cast<uint64_t>(p);
}

Result<void> device_free(uint64_t ptr) {
  BB_HIP_TRY(hipFree(reinterpret_cast<void*>(ptr)));
  return {};
}

Result<void> upload(uint64_t dst_dev, const void* src, uint64_t nbytes) {
  return copy_sync(reinterpret_cast<void*>(dst_dev), src, nbytes,
                   hipMemcpyHostToDevice);
}

Result<void> download(void* dst, uint64_t src_dev, uint64_t nbytes) {
  return copy_sync(dst, reinterpret_cast<const void*>(src_dev), nbytes,
                   hipMemcpyDeviceToHost);
}

Result<void> sync() {
  BB_HIP_TRY(hipDeviceSynchronize());
  return {};
}

// ------------------- fused copy + digest (put fast path) -------------------
namespace {

struct PutSeg {
  const uint8_t* src;
  uint8_t* dst;
  uint64_t nbytes;
};

// One desc = one whole object at object-offset 0. Waves own contiguous
// ranges of global 1-KiB tiles; each tile is loaded once with the MFMA
// A-fragment lane map (fully coalesced), stored to dst with the same map,
// and hashed in registers. Object partials flush via one atomic per
// boundary; digests finalize in a follow-up kernel.
__global__ void __launch_bounds__(kBlock)
fused_put_kernel(const PutSeg* __restrict__ descs,
                 const uint64_t* __restrict__ tile_prefix, uint32_t nobjs,
                 uint64_t total_tiles, unsigned long long* __restrict__ out) {
  using namespace blackbird::gpu::dev;
  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const uint64_t gwave =
      static_cast<uint64_t>(blockIdx.x) * (kBlock / 64) + wave;
  const uint64_t nwaves = static_cast<uint64_t>(gridDim.x) * (kBlock / 64);
  const uint64_t per = (total_tiles + nwaves - 1) / nwaves;
  const uint64_t begin = gwave * per;
  const uint64_t end =
      begin + per < total_tiles ? begin + per : total_tiles;
  if (begin >= total_tiles) return;

  const i32x4 b_frag = make_b_frag(lane);
  const WReg wr = make_w_reg(lane);
  const uint32_t lane_off = lane_tile_offset(lane);

  uint32_t oi = 0;
  {
    uint32_t lo = 0, hi = nobjs - 1;
    while (lo < hi) {
      uint32_t mid = (lo + hi + 1) >> 1;
      if (tile_prefix[mid] <= begin) lo = mid;
      else hi = mid - 1;
    }
    oi = lo;
  }
  PutSeg cur = descs[oi];
  uint64_t base_tile = tile_prefix[oi];
  uint64_t next_boundary = (oi + 1 < nobjs) ? tile_prefix[oi + 1] : ~0ull;
  uint64_t cur_full = cur.nbytes / blackbird::digest::kTileBytes;

  uint64_t h = 0;
  // software pipeline: the NEXT full tile's fragment loads while the
  // current one is stored+hashed — the VALU fold hides the load latency
  bool have_pre = false;
  i32x4 pre{};
  for (uint64_t gt = begin; gt < end; ++gt) {
    while (gt >= next_boundary) {
      h = wave_sum_u64(h);
      if (lane == 0 && h != 0) atomicAdd(&out[oi], h);
      h = 0;
      ++oi;
      cur = descs[oi];
      base_tile = next_boundary;
      next_boundary = (oi + 1 < nobjs) ? tile_prefix[oi + 1] : ~0ull;
      cur_full = cur.nbytes / blackbird::digest::kTileBytes;
      have_pre = false;  // prefetch belonged to the previous object
    }
    const uint64_t t = gt - base_tile;
    const uint64_t toff = t * blackbird::digest::kTileBytes;
    if (t < cur_full) {
      const i32x4 a =
          have_pre ? pre : load_a_frag(cur.src + toff, lane);
      const uint64_t nt = t + 1;
      have_pre = (gt + 1 < end) && (gt + 1 < next_boundary) && (nt < cur_full);
      if (have_pre)
        pre = load_a_frag(cur.src + nt * blackbird::digest::kTileBytes, lane);
      // nontemporal: an A/B test against a temporal (L2-filling) store was
      // within run variance end-to-end; the 1000-step soak record stands on
      // this version
      __builtin_nontemporal_store(
          a, reinterpret_cast<i32x4*>(cur.dst + toff + lane_off));
      h += hash_tile_frag(a, b_frag, wr, t * 64 + lane);
    } else {
      // tail tile: zero-padded hash, byte-guarded store
      const i32x4 a = load_a_frag_guarded(cur.src, toff, cur.nbytes, lane);
      const uint64_t base = toff + lane_off;
      const int8_t* ab = reinterpret_cast<const int8_t*>(&a);
#pragma unroll
      for (int j = 0; j < 16; ++j)
        if (base + j < cur.nbytes)
          cur.dst[base + j] = static_cast<uint8_t>(ab[j]);
      h += hash_tile_frag(a, b_frag, wr, t * 64 + lane);
    }
  }
  h = wave_sum_u64(h);
  if (lane == 0 && h != 0) atomicAdd(&out[oi], h);
}

__global__ void fused_put_finalize_kernel(const PutSeg* __restrict__ descs,
                                          uint32_t nobjs,
                                          unsigned long long* __restrict__ out) {
  uint32_t i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < nobjs)
    out[i] = blackbird::digest::finalize(out[i], descs[i].nbytes);
}

thread_local DescStaging g_put_stage;

}  // namespace

Result<void> fused_put(const PutDesc* descs, uint32_t n, uint64_t* out_digests,
                       hipStream_t stream) {
  if (n == 0) return {};
  const size_t segs_bytes = n * sizeof(PutSeg);
  const size_t prefix_bytes = n * sizeof(uint64_t);
  const size_t out_bytes = n * sizeof(uint64_t);
  BB_RETURN_IF_ERROR(g_put_stage.acquire(segs_bytes + prefix_bytes + out_bytes));
  auto* h_segs = static_cast<PutSeg*>(g_put_stage.pinned);
  auto* h_prefix = reinterpret_cast<uint64_t*>(
      static_cast<uint8_t*>(g_put_stage.pinned) + segs_bytes);
  auto* h_out = reinterpret_cast<uint64_t*>(
      static_cast<uint8_t*>(g_put_stage.pinned) + segs_bytes + prefix_bytes);
  auto* d_segs = static_cast<PutSeg*>(g_put_stage.device);
  auto* d_prefix = reinterpret_cast<uint64_t*>(
      static_cast<uint8_t*>(g_put_stage.device) + segs_bytes);
  auto* d_out = reinterpret_cast<unsigned long long*>(
      static_cast<uint8_t*>(g_put_stage.device) + segs_bytes + prefix_bytes);

  uint64_t total = 0;
  for (uint32_t i = 0; i < n; ++i) {
    h_segs[i] = {static_cast<const uint8_t*>(descs[i].src),
                 static_cast<uint8_t*>(descs[i].dst), descs[i].nbytes};
    h_prefix[i] = total;
    total += (descs[i].nbytes + blackbird::digest::kTileBytes - 1) /
             blackbird::digest::kTileBytes;
  }
  if (total == 0) return {};

  BB_HIP_TRY(hipMemcpyAsync(d_segs, h_segs, segs_bytes + prefix_bytes,
                            hipMemcpyHostToDevice, stream));
  BB_HIP_TRY(hipMemsetAsync(d_out, 0, out_bytes, stream));
  uint64_t waves = (total + 1) / 2;
  uint64_t blocks = std::min<uint64_t>((waves + 3) / 4, 4096);
  if (blocks < 1) blocks = 1;
  fused_put_kernel<<<static_cast<int>(blocks), kBlock, 0, stream>>>(
      d_segs, d_prefix, n, total, d_out);
  if (hipError_t _le = hipGetLastError(); _le != hipSuccess)
    return hip_error(_le, "fused_put_kernel launch");
  fused_put_finalize_kernel<<<(n + 255) / 256, 256, 0, stream>>>(d_segs, n, d_out);
  if (hipError_t _le = hipGetLastError(); _le != hipSuccess)
    return hip_error(_le, "fused_put_finalize_kernel launch");
  BB_HIP_TRY(hipMemcpyAsync(h_out, d_out, out_bytes, hipMemcpyDeviceToHost, stream));
  BB_HIP_TRY(hipStreamSynchronize(stream));
  std::memcpy(out_digests, h_out, out_bytes);
  BB_HIP_TRY(hipEventRecord(g_put_stage.ev, stream));
  return {};
}

// ----------------- hipGraph-replayed session step (FusedPutPlan) -----------
// A batch session re-runs the SAME desc list every step, so the descriptors
// are uploaded once and the step is captured as a graph: one hipGraphLaunch
// replaces memset + 2 kernel launches + D2H issue each step.

struct FusedPutPlan::Impl {
  hipStream_t stream = nullptr;   // dedicated capture/replay stream
  hipGraphExec_t exec = nullptr;
  void* dev = nullptr;            // segs + prefix + out
  uint64_t* h_out = nullptr;      // pinned digest landing zone
  size_t out_bytes = 0;

  ~Impl() {
    if (exec) (void)hipGraphExecDestroy(exec);
    if (dev) (void)hipFree(dev);
    if (h_out) (void)hipHostFree(h_out);
    if (stream) (void)hipStreamDestroy(stream);
  }
};

FusedPutPlan::~FusedPutPlan() { delete impl_; }

Result<void> FusedPutPlan::build(const PutDesc* descs, uint32_t n, int device) {
  if (impl_) return Error{ErrorCode::INVALID_ARGUMENT, "plan already built"};
  if (n == 0) return Error{ErrorCode::INVALID_ARGUMENT, "empty desc list"};
  BB_HIP_TRY(hipSetDevice(device));
  auto impl = std::make_unique<Impl>();
  const size_t segs_bytes = n * sizeof(PutSeg);
  const size_t prefix_bytes = n * sizeof(uint64_t);
  impl->out_bytes = n * sizeof(uint64_t);

  std::vector<PutSeg> h_segs(n);
  std::vector<uint64_t> h_prefix(n);
  uint64_t total = 0;
  for (uint32_t i = 0; i < n; ++i) {
    h_segs[i] = {static_cast<const uint8_t*>(descs[i].src),
                 static_cast<uint8_t*>(descs[i].dst), descs[i].nbytes};
    h_prefix[i] = total;
    total += (descs[i].nbytes + blackbird::digest::kTileBytes - 1) /
             blackbird::digest::kTileBytes;
  }
  if (total == 0) return Error{ErrorCode::INVALID_ARGUMENT, "zero-byte batch"};

  BB_HIP_TRY(hipStreamCreateWithFlags(&impl->stream, hipStreamNonBlocking));
  BB_HIP_TRY(hipMalloc(&impl->dev, segs_bytes + prefix_bytes + impl->out_bytes));
  BB_HIP_TRY(hipHostMalloc(reinterpret_cast<void**>(&impl->h_out),
                           impl->out_bytes, hipHostMallocDefault));
  auto* d_segs = static_cast<PutSeg*>(impl->dev);
  auto* d_prefix = reinterpret_cast<uint64_t*>(
      static_cast<uint8_t*>(impl->dev) + segs_bytes);
  auto* d_out = reinterpret_cast<unsigned long long*>(
      static_cast<uint8_t*>(impl->dev) + segs_bytes + prefix_bytes);
  BB_RETURN_IF_ERROR(copy_sync(d_segs, h_segs.data(), segs_bytes,
                               hipMemcpyHostToDevice));
  BB_RETURN_IF_ERROR(copy_sync(d_prefix, h_prefix.data(), prefix_bytes,
                               hipMemcpyHostToDevice));

  uint64_t waves = (total + 1) / 2;
  uint64_t blocks = std::min<uint64_t>((waves + 3) / 4, 4096);
  if (blocks < 1) blocks = 1;

  hipGraph_t graph = nullptr;
  // Relaxed: every captured op is issued on impl->stream by THIS thread, so
  // the stricter modes buy nothing — and ROCm's ThreadLocal mode still
  // fails OTHER threads' legacy-stream hipMemcpy while a capture is open
  // ("operation would make the legacy stream depend on a capturing blocking
  // stream"), which a concurrent client must be free to do.
  BB_HIP_TRY(hipStreamBeginCapture(impl->stream,
                                   hipStreamCaptureModeRelaxed));
  hipError_t cap = hipSuccess;
  do {
    if ((cap = hipMemsetAsync(d_out, 0, impl->out_bytes, impl->stream)) !=
        hipSuccess)
      break;
    fused_put_kernel<<<static_cast<int>(blocks), kBlock, 0, impl->stream>>>(
        d_segs, d_prefix, n, total, d_out);
    if ((cap = hipGetLastError()) != hipSuccess) break;
    fused_put_finalize_kernel<<<(n + 255) / 256, 256, 0, impl->stream>>>(
        d_segs, n, d_out);
    if ((cap = hipGetLastError()) != hipSuccess) break;
    cap = hipMemcpyAsync(impl->h_out, d_out, impl->out_bytes,
                         hipMemcpyDeviceToHost, impl->stream);
  } while (false);
  hipError_t endc = hipStreamEndCapture(impl->stream, &graph);
  if (cap != hipSuccess) {
    if (graph) (void)hipGraphDestroy(graph);
    return hip_error(cap, "graph capture");
  }
  BB_HIP_TRY(endc);
  hipError_t inst = hipGraphInstantiate(&impl->exec, graph, nullptr, nullptr, 0);
  (void)hipGraphDestroy(graph);
  BB_HIP_TRY(inst);
  n_ = n;
  impl_ = impl.release();
  return {};
}

Result<void> FusedPutPlan::run(uint64_t* out_digests) {
  if (!impl_) return Error{ErrorCode::INVALID_ARGUMENT, "plan not built"};
  BB_HIP_TRY(hipGraphLaunch(impl_->exec, impl_->stream));
  BB_HIP_TRY(hipStreamSynchronize(impl_->stream));
  std::memcpy(out_digests, impl_->h_out, impl_->out_bytes);
  return {};
}

}  // namespace blackbird::gpu
