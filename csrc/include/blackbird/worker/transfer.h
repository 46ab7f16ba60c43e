// Worker-side transfer engine: pull a byte range described by source shards
// into a local backend. This is the data mover behind tier migration
// (GPU→DRAM→NVMe spill and promotion).
//
// Concurrency model: a pool of independent LANES, each with its own HIP
// stream and pinned double-buffer. Concurrent DATA_PULL requests (the tier
// manager fans migrations out) each acquire a lane, so pulls neither
// serialize on a shared stream nor race on a shared bounce buffer; within
// one pull the two pinned halves ping-pong so the D2H DMA of chunk i+1
// overlaps the backend write of chunk i. Direct memcpy/SHM/IPC one-sided
// paths otherwise, TCP data protocol as the universal fallback.
#pragma once

#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <vector>

#include "blackbird/common/result.h"
#include "blackbird/common/types.h"
#include "blackbird/worker/storage_backend.h"

namespace blackbird {

class PoolMapper;
namespace rpc {
class RpcClient;
}

class TransferEngine {
 public:
  TransferEngine();
  ~TransferEngine();

  // Copy the ordered `srcs` ranges (covering [0, total) of an object) into
  // dst_backend at dst_offset. Synchronous (events gate the GPU copies).
  // Thread-safe: concurrent pulls run on independent lanes.
  Result<void> pull(StorageBackend& dst, uint64_t dst_offset,
                    const std::vector<ShardPlacement>& srcs);

 private:
  struct Lane {
    void* stream = nullptr;  // hipStream_t, created lazily
    void* pin = nullptr;     // pinned (or malloc'd without GPU) double-buffer
    bool in_use = false;
  };
  static constexpr int kLanes = 4;
  static constexpr uint64_t kChunk = 16ull << 20;  // per half

  // acquire a free lane (blocks when all kLanes are busy), release it after
  Result<Lane*> acquire_lane();
  void release_lane(Lane* l);

  Result<void> pull_one(StorageBackend& dst, uint64_t dst_offset,
                        const ShardPlacement& src, Lane& lane);
  rpc::RpcClient* data_client(const std::string& endpoint);

  std::shared_ptr<PoolMapper> mapper_;
  std::mutex mu_;
  std::map<std::string, std::unique_ptr<rpc::RpcClient>> clients_;
  std::mutex lanes_mu_;
  std::condition_variable lanes_cv_;
  Lane lanes_[kLanes];
};

}  // namespace blackbird
