#include "blackbird/worker/transfer.h"

#include <cstdlib>
#include <cstring>

#include <hip/hip_runtime_api.h>

#include "blackbird/client/pool_mapper.h"
#include "blackbird/common/trace.h"
#include "blackbird/common/log.h"
#include "blackbird/gpu/gpu_kernels.h"
#include "blackbird/rpc/methods.h"
#include "blackbird/rpc/rpc.h"

namespace blackbird {

namespace {
Error hip_err(hipError_t e, const char* what) {
  return Error{ErrorCode::HIP_ERROR,
               std::string(what) + ": " + hipGetErrorString(e)};
}
#define BB_HIP(expr)                                  \
  do {                                                \
    hipError_t _e = (expr);                           \
    if (_e != hipSuccess) return hip_err(_e, #expr);  \
  } while (0)

struct ReadReq {
  std::string pool_id;
  uint64_t offset = 0;
  uint64_t length = 0;
  BB_FIELDS(pool_id, offset, length)
};
}  // namespace

TransferEngine::TransferEngine() : mapper_(std::make_shared<PoolMapper>()) {}

TransferEngine::~TransferEngine() {
  for (auto& l : lanes_) {
    if (l.stream) (void)hipStreamDestroy(static_cast<hipStream_t>(l.stream));
    if (l.pin) {
      if (gpu::available()) (void)hipHostFree(l.pin);
      else ::free(l.pin);
    }
  }
}

Result<TransferEngine::Lane*> TransferEngine::acquire_lane() {
  std::unique_lock<std::mutex> lk(lanes_mu_);
  Lane* l = nullptr;
  lanes_cv_.wait(lk, [&] {
    for (auto& cand : lanes_)
      if (!cand.in_use) {
        l = &cand;
        return true;
      }
    return false;
  });
  l->in_use = true;
  lk.unlock();
  // lazy resource creation outside the pool lock (the lane is ours)
  if (!l->pin) {
    if (gpu::available()) {
      void* p = nullptr;
      if (hipHostMalloc(&p, 2 * kChunk, hipHostMallocDefault) != hipSuccess) {
        release_lane(l);
        return Error{ErrorCode::HIP_ERROR, "lane staging alloc"};
      }
      l->pin = p;
    } else {
      l->pin = ::malloc(2 * kChunk);
      if (!l->pin) {
        release_lane(l);
        return Error{ErrorCode::INTERNAL_ERROR, "lane staging alloc"};
      }
    }
  }
  if (!l->stream && gpu::available()) {
    hipStream_t s = nullptr;
    if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) != hipSuccess) {
      release_lane(l);
      return Error{ErrorCode::HIP_ERROR, "lane stream create"};
    }
    l->stream = s;
  }
  return l;
}

void TransferEngine::release_lane(Lane* l) {
  {
    std::lock_guard<std::mutex> g(lanes_mu_);
    l->in_use = false;
  }
  lanes_cv_.notify_one();
}

rpc::RpcClient* TransferEngine::data_client(const std::string& endpoint) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = clients_.find(endpoint);
  if (it != clients_.end()) return it->second.get();
  auto c = std::make_unique<rpc::RpcClient>();
  if (!c->connect(endpoint).ok()) return nullptr;
  return clients_.emplace(endpoint, std::move(c)).first->second.get();
}

Result<void> TransferEngine::pull(StorageBackend& dst, uint64_t dst_offset,
                                  const std::vector<ShardPlacement>& srcs) {
  BB_TRACE_SCOPE("bb::tier_pull");
  auto lane = acquire_lane();
  if (!lane.ok()) return lane.error();
  uint64_t off = dst_offset;
  Result<void> rc{};
  for (const auto& s : srcs) {
    rc = pull_one(dst, off, s, *lane.value());
    if (!rc.ok()) break;
    off += s.length;
  }
  release_lane(lane.value());
  return rc;
}

Result<void> TransferEngine::pull_one(StorageBackend& dst, uint64_t dst_offset,
                                      const ShardPlacement& src, Lane& lane) {
  const bool dst_is_gpu = dst.storage_class() == StorageClass::RAM_GPU;
  uint8_t* dst_ptr = static_cast<uint8_t*>(dst.base_ptr());

  // Bound the destination range against the pool BEFORE any transfer: the
  // request arrives over the network-exposed data port, and the raw-pointer
  // paths below bypass the backend's own check_range.
  if (src.length > dst.capacity() || dst_offset > dst.capacity() - src.length)
    return Error{ErrorCode::INVALID_OFFSET, "pull dst range out of bounds"};

  // resolve source to a pointer when one-sided access is possible; when the
  // mapping size is known, reject out-of-range source slices the same way
  const uint8_t* src_ptr = nullptr;
  bool src_is_gpu = false;
  {
    bool is_dev = false;
    uint64_t src_pool_size = 0;  // 0 = unknown (IPC import)
    if (void* base = LocalPools::inst().lookup(src.pool_id, &is_dev, nullptr,
                                               &src_pool_size)) {
      src_ptr = static_cast<const uint8_t*>(base) + src.offset;
      src_is_gpu = is_dev;
    } else if (src.access.kind == AccessKind::SHM && !src.access.shm_name.empty()) {
      if (void* base =
              mapper_->map_shm(src.access.shm_name, 0, &src_pool_size))
        src_ptr = static_cast<const uint8_t*>(base) + src.offset;
    } else if (src.access.kind == AccessKind::HIP_IPC &&
               !src.access.ipc_handle_hex.empty() && gpu::available()) {
      if (void* base =
              mapper_->open_ipc(src.access.ipc_handle_hex, src.access.device_id)) {
        src_ptr = static_cast<const uint8_t*>(base) + src.offset;
        src_is_gpu = true;
      }
    }
    if (src_ptr && src_pool_size > 0 &&
        (src.length > src_pool_size ||
         src.offset > src_pool_size - src.length))
      return Error{ErrorCode::INVALID_OFFSET, "pull src range out of bounds"};
  }

  if (src_ptr && !dst_ptr) {
    // destination has no memory mapping (direct-IO file tier): move through
    // backend write(). Host sources write straight through; GPU sources
    // ping-pong the lane's two pinned halves so the D2H DMA of chunk i+1
    // overlaps the (CPU/NVMe) backend write of chunk i.
    if (!src_is_gpu) return dst.write(dst_offset, src_ptr, src.length);
    auto stream = static_cast<hipStream_t>(lane.stream);
    uint8_t* half[2] = {static_cast<uint8_t*>(lane.pin),
                        static_cast<uint8_t*>(lane.pin) + kChunk};
    const uint64_t n_chunks = (src.length + kChunk - 1) / kChunk;
    uint64_t cur_len = std::min(src.length, kChunk);
    BB_HIP(hipMemcpyAsync(half[0], src_ptr, cur_len, hipMemcpyDeviceToHost,
                          stream));
    BB_HIP(hipStreamSynchronize(stream));
    for (uint64_t c = 0; c < n_chunks; ++c) {
      const int cur = static_cast<int>(c & 1);
      uint64_t next_off = (c + 1) * kChunk;
      uint64_t next_len = 0;
      if (c + 1 < n_chunks) {
        next_len = std::min(src.length - next_off, kChunk);
        BB_HIP(hipMemcpyAsync(half[1 - cur], src_ptr + next_off, next_len,
                              hipMemcpyDeviceToHost, stream));
      }
      BB_RETURN_IF_ERROR(dst.write(dst_offset + c * kChunk, half[cur], cur_len));
      if (c + 1 < n_chunks) {
        BB_HIP(hipStreamSynchronize(stream));
        cur_len = next_len;
      }
    }
    return {};
  }
  if (src_ptr && dst_ptr) {
    if (!src_is_gpu && !dst_is_gpu) {
      std::memcpy(dst_ptr + dst_offset, src_ptr, src.length);
      return {};
    }
    // at least one GPU endpoint: hipMemcpyAsync on the LANE's stream
    // (PINNED_CPU destinations take the DMA fast path); concurrent pulls
    // ride different lanes/streams, so the engines overlap
    auto stream = static_cast<hipStream_t>(lane.stream);
    hipMemcpyKind kind = src_is_gpu
                             ? (dst_is_gpu ? hipMemcpyDeviceToDevice
                                           : hipMemcpyDeviceToHost)
                             : hipMemcpyHostToDevice;
    BB_HIP(hipMemcpyAsync(dst_ptr + dst_offset, src_ptr, src.length, kind, stream));
    BB_HIP(hipStreamSynchronize(stream));
    return {};
  }

  // TCP fallback: read from the source worker's data plane into the lane's
  // staging halves, then backend write (handles GPU destinations internally).
  auto* dc = data_client(src.access.endpoint);
  if (!dc)
    return Error{ErrorCode::CONNECT_FAILED,
                 "pull: data plane " + src.access.endpoint};
  uint64_t done = 0;
  while (done < src.length) {
    uint64_t chunk = std::min(src.length - done, 2 * kChunk);
    ReadReq req{src.pool_id, src.offset + done, chunk};
    auto r = dc->call_raw(rpc::methods::DATA_READ, serde::to_bytes(req));
    if (!r.ok()) return r.error();
    if (r.value().size() != chunk)
      return Error{ErrorCode::SIZE_MISMATCH, "pull short read"};
    BB_RETURN_IF_ERROR(dst.write(dst_offset + done, r.value().data(), chunk));
    done += chunk;
  }
  return {};
}

}  // namespace blackbird
