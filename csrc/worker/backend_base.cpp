#include "blackbird/worker/storage_backend.h"

namespace blackbird {

BackendBase::BackendBase(uint64_t capacity, uint64_t reservation_ttl_ms,
                         uint64_t alignment)
    : capacity_(capacity),
      reservation_ttl_ms_(reservation_ttl_ms),
      alloc_(capacity, PoolAllocator::Policy::BEST_FIT, alignment) {}

Result<void> BackendBase::check_range(uint64_t offset, uint64_t len) const {
  if (offset + len > capacity_ || offset + len < offset)
    return Error{ErrorCode::INVALID_OFFSET,
                 "range [" + std::to_string(offset) + ", +" + std::to_string(len) +
                     ") beyond capacity " + std::to_string(capacity_)};
  return {};
}

void BackendBase::expire_reservations_locked(uint64_t now) {
  for (auto it = reservations_.begin(); it != reservations_.end();) {
    if (it->second.expires_ms <= now) {
      alloc_.free(it->second.offset, it->second.size);
      it = reservations_.erase(it);
    } else {
      ++it;
    }
  }
}

Result<ReservationToken> BackendBase::reserve(uint64_t size) {
  auto off = alloc_.allocate(size);
  if (!off.ok()) return off.error();
  ReservationToken t;
  t.token_id = next_token_++;
  t.offset = off.value();
  t.size = size;
  t.expires_ms = now_ms() + reservation_ttl_ms_;
  std::lock_guard<std::mutex> g(mu_);
  expire_reservations_locked(now_ms());
  reservations_[t.token_id] = t;
  return t;
}

Result<ReservationToken> BackendBase::reserve_at(uint64_t offset, uint64_t size) {
  BB_RETURN_IF_ERROR(check_range(offset, size));
  BB_RETURN_IF_ERROR(alloc_.reserve_exact(offset, size));
  ReservationToken t;
  t.token_id = next_token_++;
  t.offset = offset;
  t.size = size;
  t.expires_ms = now_ms() + reservation_ttl_ms_;
  std::lock_guard<std::mutex> g(mu_);
  reservations_[t.token_id] = t;
  return t;
}

Result<void> BackendBase::commit(uint64_t token_id) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = reservations_.find(token_id);
  if (it == reservations_.end())
    return Error{ErrorCode::RESERVATION_NOT_FOUND, std::to_string(token_id)};
  if (it->second.expires_ms <= now_ms()) {
    alloc_.free(it->second.offset, it->second.size);
    reservations_.erase(it);
    return Error{ErrorCode::RESERVATION_EXPIRED, std::to_string(token_id)};
  }
  shards_[it->second.offset] = it->second.size;
  reservations_.erase(it);
  return {};
}

Result<void> BackendBase::abort(uint64_t token_id) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = reservations_.find(token_id);
  if (it == reservations_.end())
    return Error{ErrorCode::RESERVATION_NOT_FOUND, std::to_string(token_id)};
  alloc_.free(it->second.offset, it->second.size);
  reservations_.erase(it);
  return {};
}

Result<void> BackendBase::free(uint64_t offset, uint64_t size) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = shards_.find(offset);
  if (it == shards_.end())
    return Error{ErrorCode::SHARD_NOT_FOUND, std::to_string(offset)};
  if (it->second != size)
    return Error{ErrorCode::SIZE_MISMATCH,
                 "shard@" + std::to_string(offset) + " is " +
                     std::to_string(it->second) + ", not " + std::to_string(size)};
  BB_RETURN_IF_ERROR(alloc_.free(offset, size));
  shards_.erase(it);
  return {};
}

StorageStats BackendBase::stats() const {
  std::lock_guard<std::mutex> g(mu_);
  StorageStats s;
  s.capacity = capacity_;
  s.used = alloc_.used();
  s.num_shards = shards_.size();
  s.num_reservations = reservations_.size();
  for (const auto& [id, t] : reservations_) s.reserved += t.size;
  return s;
}

}  // namespace blackbird

namespace blackbird {

LocalPools& LocalPools::inst() {
  static LocalPools g;
  return g;
}

void LocalPools::add(const PoolId& id, void* base, uint64_t size,
                     bool is_device, int device, StorageBackend* backend) {
  std::lock_guard<std::mutex> g(mu_);
  pools_[id] = Entry{base, size, is_device, device, backend};
}

StorageBackend* LocalPools::backend(const PoolId& id) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = pools_.find(id);
  return it == pools_.end() ? nullptr : it->second.backend;
}

void LocalPools::remove(const PoolId& id) {
  std::lock_guard<std::mutex> g(mu_);
  pools_.erase(id);
}

void* LocalPools::lookup(const PoolId& id, bool* is_device, int* device,
                         uint64_t* size) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = pools_.find(id);
  if (it == pools_.end()) return nullptr;
  if (is_device) *is_device = it->second.is_device;
  if (device) *device = it->second.device;
  if (size) *size = it->second.size;
  return it->second.base;
}

}  // namespace blackbird
