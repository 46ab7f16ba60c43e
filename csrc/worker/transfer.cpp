#include "blackbird/worker/transfer.h"

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
  if (side_stream_) (void)hipStreamDestroy(static_cast<hipStream_t>(side_stream_));
  if (staging_) (void)hipHostFree(staging_);
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
  uint64_t off = dst_offset;
  for (const auto& s : srcs) {
    BB_RETURN_IF_ERROR(pull_one(dst, off, s));
    off += s.length;
  }
  return {};
}

Result<void> TransferEngine::pull_one(StorageBackend& dst, uint64_t dst_offset,
                                      const ShardPlacement& src) {
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
    // backend write(); GPU sources stage via a pinned bounce first
    if (!src_is_gpu) return dst.write(dst_offset, src_ptr, src.length);
    if (!staging_) {
      BB_HIP(hipHostMalloc(&staging_, staging_size_, hipHostMallocDefault));
    }
    uint64_t done = 0;
    while (done < src.length) {
      uint64_t chunk = std::min(src.length - done, staging_size_);
      BB_HIP(hipMemcpy(staging_, src_ptr + done, chunk, hipMemcpyDeviceToHost));
      BB_RETURN_IF_ERROR(dst.write(dst_offset + done, staging_, chunk));
      done += chunk;
    }
    return {};
  }
  if (src_ptr && dst_ptr) {
    if (!src_is_gpu && !dst_is_gpu) {
      std::memcpy(dst_ptr + dst_offset, src_ptr, src.length);
      return {};
    }
    // at least one GPU endpoint: hipMemcpyAsync on the side stream, gated by
    // stream sync (PINNED_CPU destinations take the DMA fast path).
    if (!side_stream_) {
      hipStream_t s = nullptr;
      BB_HIP(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
      side_stream_ = s;
    }
    auto stream = static_cast<hipStream_t>(side_stream_);
    hipMemcpyKind kind = src_is_gpu
                             ? (dst_is_gpu ? hipMemcpyDeviceToDevice
                                           : hipMemcpyDeviceToHost)
                             : hipMemcpyHostToDevice;
    BB_HIP(hipMemcpyAsync(dst_ptr + dst_offset, src_ptr, src.length, kind, stream));
    BB_HIP(hipStreamSynchronize(stream));
    return {};
  }

  // TCP fallback: read from the source worker's data plane into staging,
  // then backend write (handles GPU destinations internally).
  auto* dc = data_client(src.access.endpoint);
  if (!dc)
    return Error{ErrorCode::CONNECT_FAILED,
                 "pull: data plane " + src.access.endpoint};
  if (!staging_) {
    if (gpu::available()) {
      BB_HIP(hipHostMalloc(&staging_, staging_size_, hipHostMallocDefault));
    } else {
      staging_ = malloc(staging_size_);
      if (!staging_) return Error{ErrorCode::INTERNAL_ERROR, "staging alloc"};
    }
  }
  uint64_t done = 0;
  while (done < src.length) {
    uint64_t chunk = std::min(src.length - done, staging_size_);
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
