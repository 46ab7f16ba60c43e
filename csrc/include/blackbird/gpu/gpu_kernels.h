// Host-callable entry points for the CDNA4 (gfx950) kernels of the HBM tier.
// All functions are implemented in csrc/hip/*.hip, compiled for gfx950 only.
// They fail loudly (Result error) when no GPU is present — no silent CPU
// fallback on a GPU box.
#pragma once

#include <cstdint>

#include "blackbird/common/result.h"

// Forward decl to avoid pulling hip_runtime into every TU.
struct ihipStream_t;
typedef struct ihipStream_t* hipStream_t;

namespace blackbird::gpu {

bool available();        // HIP runtime + ≥1 device
int device_count();

// ------------------------------------------------------------- checksum
// 64-bit object digest computed on-device with MFMA i8 matrix products:
// data is consumed as 32×32 int8 tiles A_t; each tile is projected through a
// fixed pseudo-random matrix B (mfma_i32_32x32x32_i8), the 32×32 i32 product
// is folded to a u64, weighted by splitmix64(tile_index) and summed (u64
// wraparound) — position-sensitive, order-independent (parallel-friendly),
// bitwise reproducible on CPU (gpu_checksum_cpu). Tail bytes (<1 KiB) are
// hashed on the fly inside the kernel.
//
// dev_ptr: device pointer; stream: HIP stream (nullptr = default).
// Result digest is written to *out (host) after stream sync by the _sync
// variant; the async variant writes into a device/pinned u64.
Result<uint64_t> checksum_sync(const void* dev_ptr, uint64_t nbytes, int device,
                               hipStream_t stream);
Result<void> checksum_async(const void* dev_ptr, uint64_t nbytes,
                            uint64_t* dev_out, hipStream_t stream);

// Exact CPU reference of the same digest (used for verification and for
// DRAM/disk-tier objects).
uint64_t checksum_cpu(const void* ptr, uint64_t nbytes);

// Streaming CPU digest: hash `nbytes` starting at tile index `first_tile`
// (chunk must start on a 1 KiB tile boundary of the object; the final chunk
// may end mid-tile — zero-padded per spec). Combine partials with +, then
// checksum_cpu_finalize(H, object_size).
uint64_t checksum_cpu_tiles(const void* chunk, uint64_t nbytes,
                            uint64_t first_tile);
uint64_t checksum_cpu_finalize(uint64_t h, uint64_t object_nbytes);

// Batched digest: one launch hashes n device buffers; out_digests is a HOST
// array of n results (call blocks on `stream`).
Result<void> checksum_batch(const void* const* dev_ptrs, const uint64_t* sizes,
                            uint32_t n, uint64_t* out_digests, int device,
                            hipStream_t stream);

// Single 32×32×32 i8 MFMA with the assumed fragment layout (layout
// verification test). host_a/host_b: 1024 int8 row-major; host_c: 1024 i32.
Result<void> mfma_i8_probe(const int8_t* host_a, const int8_t* host_b,
                           int32_t* host_c, int device);

// ------------------------------------------- batched scatter/gather copy
// One launch serves a whole batch of object transfers (the fused multi-object
// path behind batch_put/batch_get). Descriptors are copied to device memory
// by the caller.
struct CopyDesc {
  const void* src;
  void* dst;
  uint64_t nbytes;
};
// descs: HOST array of n descriptors (the implementation stages them).
Result<void> batched_copy(const CopyDesc* descs, uint32_t n, hipStream_t stream);

// Fused put: copy whole single-shard objects (src→dst, both device-local,
// 16B-aligned) AND compute their bbhash64 digests in ONE kernel — the data
// is read once instead of twice (copy + hash). out_digests: HOST array
// (call blocks on `stream`).
struct PutDesc {
  const void* src;
  void* dst;
  uint64_t nbytes;
};
Result<void> fused_put(const PutDesc* descs, uint32_t n, uint64_t* out_digests,
                       hipStream_t stream);

// Pre-instantiated fused-put step for batch sessions (fixed desc list every
// step): descriptors live on-device (uploaded once by build), and the whole
// step — digest reset → fused copy+digest kernel → finalize → digest D2H —
// is captured in a hipGraph and replayed as ONE graph launch per step
// instead of 5 stream ops. run() blocks until the digests are in
// out_digests[n].
class FusedPutPlan {
 public:
  FusedPutPlan() = default;
  ~FusedPutPlan();
  FusedPutPlan(const FusedPutPlan&) = delete;
  FusedPutPlan& operator=(const FusedPutPlan&) = delete;

  Result<void> build(const PutDesc* descs, uint32_t n, int device);
  bool ready() const { return impl_ != nullptr; }
  uint32_t size() const { return n_; }
  Result<void> run(uint64_t* out_digests);

 private:
  struct Impl;
  Impl* impl_ = nullptr;
  uint32_t n_ = 0;
};

// Simple utilities used by tests/benchmarks.
Result<void> fill_pattern(void* dev_ptr, uint64_t nbytes, uint64_t seed,
                          hipStream_t stream);
Result<uint64_t> verify_pattern(const void* dev_ptr, uint64_t nbytes, uint64_t seed,
                                hipStream_t stream);  // returns #mismatched u64

// Synchronous copy on a per-thread NON-BLOCKING stream. The library never
// issues work on the legacy (null) stream: ROCm fails legacy-stream ops in
// EVERY thread while any hipGraph capture is open (sessions capture their
// step graphs lazily), so all blocking copies route through here. kind is a
// hipMemcpyKind (int to keep hip types out of this header).
Result<void> copy_sync(void* dst, const void* src, uint64_t nbytes, int kind);

// Raw device memory helpers (Python test harness / bench plumbing).
Result<uint64_t> device_malloc(uint64_t nbytes, int device);
Result<void> device_free(uint64_t ptr);
Result<void> upload(uint64_t dst_dev, const void* src, uint64_t nbytes);
Result<void> download(void* dst, uint64_t src_dev, uint64_t nbytes);
Result<void> sync();

}  // namespace blackbird::gpu
