// Collective batch shuffle — the feature-store all-to-all pattern
// (BASELINE config #5 at N ranks): every rank wants a set of objects owned
// by each peer, each step. Instead of N·B independent one-sided reads over
// IPC, the ranks cooperate: want-lists are exchanged, every owner gathers
// the requested objects out of its LOCAL pool with one batched-copy launch,
// and the payloads move in ONE grouped all-to-all-v (RCCL over xGMI — all 7
// links of every GPU active at once).
//
// This is the RCCL-integrated analogue of the reference's client-side
// parallel shard fan-out (blackbird_client.cpp:252-267), re-expressed as a
// collective. The per-shard one-sided IPC path stays the default transport;
// bench.py A/Bs the two on multi-GPU runs.
//
// The algorithm is written against two tiny interfaces (Exchanger = the
// collective byte transport, Copier = batched memory ops) so the planning,
// encoding, validation and layout logic runs under plain host memcpy in CPU
// tests; RcclExchanger/GpuCopier are the production implementations.
#pragma once

#include <cstdint>
#include <functional>
#include <memory>
#include <string>
#include <vector>

#include "blackbird/common/result.h"
#include "blackbird/common/types.h"
#include "blackbird/gpu/gpu_kernels.h"

namespace blackbird {

class RcclEngine;

// Keys this rank wants FROM one peer; received contiguously (in key order)
// at recv_base. sizes[] are the EXACT object sizes (the owner validates and
// fails the exchange on mismatch — the pattern is fixed-size features).
struct ShuffleWant {
  std::vector<ObjectKey> keys;
  std::vector<uint64_t> sizes;
  void* recv_base = nullptr;
};

// Collective byte transport: segment i goes to / comes from rank i. The self
// slot is served by the implementation (local copy).
class Exchanger {
 public:
  virtual ~Exchanger() = default;
  virtual int rank() const = 0;
  virtual int nranks() const = 0;
  virtual Result<void> alltoallv(const std::vector<const void*>& send_ptrs,
                                 const std::vector<uint64_t>& send_bytes,
                                 const std::vector<void*>& recv_ptrs,
                                 const std::vector<uint64_t>& recv_bytes) = 0;
};

// Batched memory ops over the exchange buffers (GPU kernels or host memcpy).
class Copier {
 public:
  virtual ~Copier() = default;
  virtual Result<void*> alloc(uint64_t nbytes) = 0;
  virtual void free(void* p) = 0;
  virtual Result<void> copy_batch(const gpu::CopyDesc* descs, uint32_t n) = 0;
  // exchange metadata (want-lists) lives on the transport's memory space:
  // host for the loopback exchanger, device for RCCL
  virtual Result<void> to_transport(void* dst, const void* host_src,
                                    uint64_t n) = 0;
  virtual Result<void> from_transport(void* host_dst, const void* src,
                                      uint64_t n) = 0;
};

// Resolve an object key this rank OWNS to a transport-visible pointer of
// exactly `size` bytes (nullptr = unknown/size mismatch → exchange fails).
using ShuffleResolver =
    std::function<const void*(const ObjectKey& key, uint64_t size)>;

// The collective: every rank calls with its per-peer want lists
// (want.size() == nranks; the self slot is served locally). Blocking.
Result<void> batch_shuffle(Exchanger& ex, Copier& cp,
                           const ShuffleResolver& resolve,
                           const std::vector<ShuffleWant>& want);

// ---- production implementations ----

// RCCL-backed exchanger: one grouped send/recv per call, on `stream`,
// synchronized before returning (collective completion = data visible).
class RcclExchanger : public Exchanger {
 public:
  RcclExchanger(RcclEngine& e, hipStream_t stream) : e_(e), stream_(stream) {}
  int rank() const override;
  int nranks() const override;
  Result<void> alltoallv(const std::vector<const void*>& sp,
                         const std::vector<uint64_t>& sb,
                         const std::vector<void*>& rp,
                         const std::vector<uint64_t>& rb) override;

 private:
  RcclEngine& e_;
  hipStream_t stream_;
};

// Device copier: hipMalloc staging + the fused batched_copy kernel.
class GpuCopier : public Copier {
 public:
  GpuCopier(int device, hipStream_t stream) : device_(device), stream_(stream) {}
  Result<void*> alloc(uint64_t nbytes) override;
  void free(void* p) override;
  Result<void> copy_batch(const gpu::CopyDesc* descs, uint32_t n) override;
  Result<void> to_transport(void* dst, const void* src, uint64_t n) override;
  Result<void> from_transport(void* dst, const void* src, uint64_t n) override;

 private:
  int device_;
  hipStream_t stream_;
};

// ---- host loopback (tests / CPU-tier shuffles) ----
// N ranks in one process (threads) exchanging through shared host memory.
// Mirrors the collective semantics: alltoallv blocks until every rank of the
// group posted its segments.
class LoopbackGroup {
 public:
  explicit LoopbackGroup(int nranks);
  ~LoopbackGroup();
  int nranks() const { return nranks_; }

  class Rank;
  // exchanger for one rank (callable from its own thread)
  std::unique_ptr<Exchanger> exchanger(int rank);

 private:
  friend class Rank;
  struct State;
  int nranks_;
  std::shared_ptr<State> st_;
};

// Host copier (memcpy; transport memory = host memory).
class HostCopier : public Copier {
 public:
  Result<void*> alloc(uint64_t nbytes) override;
  void free(void* p) override;
  Result<void> copy_batch(const gpu::CopyDesc* descs, uint32_t n) override;
  Result<void> to_transport(void* dst, const void* src, uint64_t n) override;
  Result<void> from_transport(void* dst, const void* src, uint64_t n) override;
};

}  // namespace blackbird
