#include "blackbird/keystone/keystone_service.h"

#include <algorithm>
#include <atomic>
#include <random>
#include <thread>

#include "blackbird/common/log.h"
#include "blackbird/rpc/methods.h"

namespace blackbird {

namespace {
// relaxed atomic load of a hot ObjectMeta field for SHARED-lock readers
// (session commits mutate these fields under the shared lock)
template <typename T>
inline T rload(const T& f) {
  return std::atomic_ref<T>(const_cast<T&>(f)).load(std::memory_order_relaxed);
}

}  // namespace

namespace {
std::string random_id() {
  static std::mt19937_64 rng(std::random_device{}());
  char buf[20];
  snprintf(buf, sizeof(buf), "%016llx", (unsigned long long)rng());
  return buf;
}
}  // namespace

KeystoneService::KeystoneService(KeystoneConfig config,
                                 std::shared_ptr<coord::CoordService> coord)
    : config_(std::move(config)), coord_(std::move(coord)),
      instance_id_("keystone-" + random_id()) {
  if (!coord_) coord_ = coord::make_coord(config_.coord_endpoint);
}

KeystoneService::~KeystoneService() { stop(); }

Result<void> KeystoneService::initialize() {
  if (!coord_) return Error{ErrorCode::COORD_UNAVAILABLE, "no coordination service"};
  return {};
}

Result<void> KeystoneService::start() {
  if (running_.exchange(true)) return {};
  if (auto* cc = dynamic_cast<coord::CoordClient*>(coord_.get())) {
    cc->set_on_reconnect([this] {
      if (!running_.load()) return;  // teardown already under way
      // the (in-memory) coordination server restarted: workers re-register
      // themselves; re-scan to pick their state up promptly
      BB_LOG(WARN) << "coordination restarted — rescanning cluster state";
      load_existing_state();
    });
  }
  load_existing_state();
  setup_watchers();
  if (config_.enable_ha) {
    elector_ = std::make_unique<coord::LeaderElector>(
        coord_, prefix() + "/leader", instance_id_, config_.worker_ttl_ms);
    elector_->set_on_elected([this] {
      if (!running_.load()) return;
      // a promoted standby's in-memory maps are stale: rescan workers,
      // pools and (with persist_objects) the persisted object map
      BB_LOG(WARN) << "promoted to keystone leader — rescanning state";
      load_existing_state();
    });
    elector_->start();
  }
  gc_thread_ = std::thread([this] { gc_loop(); });
  keepalive_thread_ = std::thread([this] { keepalive_loop(); });
  if (config_.persist_objects)
    persist_thread_ = std::thread([this] { persist_loop(); });
  BB_LOG(INFO) << "keystone started (cluster " << config_.cluster_id << ", "
               << instance_id_ << ")";
  return {};
}

void KeystoneService::stop() {
  if (!running_.exchange(false)) return;
  {
    std::lock_guard<std::mutex> g(cv_mu_);  // no lost wakeup on stop
  }
  cv_.notify_all();
  if (gc_thread_.joinable()) gc_thread_.join();
  if (keepalive_thread_.joinable()) keepalive_thread_.join();
  if (persist_thread_.joinable()) persist_thread_.join();
  if (elector_) elector_->stop();
  for (auto id : watch_ids_) coord_->unwatch(id);
  watch_ids_.clear();
  // a watch callback may still be mid-flight on the coordination client's
  // dispatcher thread (e.g. a heartbeat expiry sweeping thousands of dead
  // copies) — wait it out before the caller may destroy this object
  while (cb_inflight_.load() > 0)
    std::this_thread::sleep_for(std::chrono::milliseconds(1));
  coord_->del("/blackbird/services/blackbird-keystone/" + instance_id_);
}

bool KeystoneService::is_leader() const {
  return !elector_ || elector_->is_leader();
}

// ------------------------------------------------------------- object ops

bool KeystoneService::object_exists(const ObjectKey& key) {
  std::shared_lock lk(objects_mu_);
  auto it = objects_.find(key);
  if (it == objects_.end()) return false;
  const ObjectMeta& m = it->second;
  if (rload(m.state) != ObjectState::COMMITTED) return false;
  return !(m.ttl_ms > 0 && now_ms() > rload(m.created_ms) + m.ttl_ms);
}

Result<GetWorkersResponse> KeystoneService::get_workers(const ObjectKey& key) {
  std::unique_lock lk(objects_mu_);
  auto it = objects_.find(key);
  if (it == objects_.end()) return Error{ErrorCode::OBJECT_NOT_FOUND, key};
  auto& meta = it->second;
  if (meta.state != ObjectState::COMMITTED)
    return Error{ErrorCode::OBJECT_NOT_COMMITTED, key};
  if (meta.expired(now_ms())) {
    remove_object_locked(key);
    return Error{ErrorCode::OBJECT_EXPIRED, key};
  }
  meta.last_access_ms = now_ms();
  meta.access_count++;
  GetWorkersResponse resp;
  resp.copies = meta.copies;
  resp.size = meta.size;
  resp.checksum = meta.checksum;
  return resp;
}

Result<PutStartResponse> KeystoneService::put_start(const ObjectKey& key,
                                                    uint64_t size,
                                                    const PlacementConfig& cfg) {
  if (!is_leader())
    return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
  if (key.empty()) return Error{ErrorCode::INVALID_ARGUMENT, "empty key"};
  if (size == 0) return Error{ErrorCode::INVALID_ARGUMENT, "zero-size object"};
  std::unique_lock lk(objects_mu_);
  auto it = objects_.find(key);
  if (it != objects_.end()) {
    if (!it->second.expired(now_ms()) && !cfg.replace)
      return Error{ErrorCode::OBJECT_EXISTS, key};
    ObjectMeta& m = it->second;
    if (cfg.replace && m.state == ObjectState::COMMITTED && m.size == size &&
        m.copies.size() == std::max<uint32_t>(cfg.replication, 1)) {
      // same-size upsert: overwrite in place (placements stay stable)
      m.state = ObjectState::PENDING;
      m.checksum = 0;
      PutStartResponse resp;
      resp.copies = m.copies;
      resp.view_version = view_version_.load();
      return resp;
    }
    remove_object_locked(key);
  }
  auto placed = allocator_.allocate(key, size, cfg);
  if (!placed.ok()) return placed.error();

  ObjectMeta meta;
  meta.key = key;
  meta.size = size;
  meta.ttl_ms = cfg.ttl_ms ? cfg.ttl_ms : config_.object_ttl_default_ms;
  meta.created_ms = now_ms();
  meta.last_access_ms = meta.created_ms;
  meta.state = ObjectState::PENDING;
  meta.replication = std::max<uint32_t>(cfg.replication, 1);
  meta.copies = placed.value();
  objects_[key] = meta;
  bump_view();

  PutStartResponse resp;
  resp.copies = std::move(placed.value());
  resp.view_version = view_version_.load();
  return resp;
}

namespace {
// Record per-shard digests into the copies; single-shard copies default to
// the whole-object checksum (their shard IS the object).
void apply_shard_digests(ObjectMeta& m, uint64_t checksum,
                         const std::vector<std::vector<uint64_t>>& sd) {
  for (size_t c = 0; c < m.copies.size(); ++c) {
    auto& shards = m.copies[c].shards;
    if (c < sd.size() && sd[c].size() == shards.size()) {
      for (size_t s = 0; s < shards.size(); ++s) shards[s].digest = sd[c][s];
    } else if (shards.size() == 1) {
      shards[0].digest = checksum;
    }
  }
}

// shared-lock variant: digest slots may be read concurrently by get batches
void apply_shard_digests_atomic(ObjectMeta& m, uint64_t checksum,
                                const std::vector<std::vector<uint64_t>>& sd) {
  for (size_t c = 0; c < m.copies.size(); ++c) {
    auto& shards = m.copies[c].shards;
    if (c < sd.size() && sd[c].size() == shards.size()) {
      for (size_t s = 0; s < shards.size(); ++s)
        std::atomic_ref<uint64_t>(shards[s].digest)
            .store(sd[c][s], std::memory_order_relaxed);
    } else if (shards.size() == 1) {
      std::atomic_ref<uint64_t>(shards[0].digest)
          .store(checksum, std::memory_order_relaxed);
    }
  }
}
}  // namespace

Result<void> KeystoneService::put_complete(
    const ObjectKey& key, uint64_t checksum,
    const std::vector<std::vector<uint64_t>>& shard_digests) {
  std::unique_lock lk(objects_mu_);
  auto it = objects_.find(key);
  if (it == objects_.end()) return Error{ErrorCode::OBJECT_NOT_FOUND, key};
  if (it->second.state == ObjectState::COMMITTED) {
    if (it->second.checksum == checksum) return {};  // retried commit
    return Error{ErrorCode::INVALID_STATE, "already committed: " + key};
  }
  it->second.state = ObjectState::COMMITTED;
  it->second.checksum = checksum;
  apply_shard_digests(it->second, checksum, shard_digests);
  it->second.created_ms = now_ms();  // TTL starts at commit
  it->second.last_access_ms = it->second.created_ms;
  mark_dirty_locked(key, false);
  bump_view();
  lk.unlock();
  flush_dirty_now();
  return {};
}

Result<void> KeystoneService::put_cancel(const ObjectKey& key) {
  std::unique_lock lk(objects_mu_);
  auto it = objects_.find(key);
  if (it == objects_.end()) return Error{ErrorCode::OBJECT_NOT_FOUND, key};
  if (it->second.state == ObjectState::COMMITTED)
    return Error{ErrorCode::INVALID_STATE, "committed; use remove_object"};
  return remove_object_locked(key);
}

Result<void> KeystoneService::remove_object_locked(const ObjectKey& key) {
  allocator_.free(key);
  objects_.erase(key);
  bump_placement_epoch_locked();  // session meta pointers may now dangle
  mark_dirty_locked(key, true);
  bump_view();
  return {};
}

Result<void> KeystoneService::remove_object(const ObjectKey& key) {
  if (!is_leader())
    return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
  std::unique_lock lk(objects_mu_);
  if (!objects_.count(key)) return Error{ErrorCode::OBJECT_NOT_FOUND, key};
  return remove_object_locked(key);
}

std::vector<ObjectSummary> KeystoneService::list_objects(
    const std::string& prefix, uint32_t limit) {
  std::vector<ObjectSummary> out;
  const uint64_t now = now_ms();
  std::shared_lock lk(objects_mu_);
  for (const auto& [key, meta] : objects_) {
    if (out.size() >= limit) break;
    if (rload(meta.state) != ObjectState::COMMITTED ||
        (meta.ttl_ms > 0 && now > rload(meta.created_ms) + meta.ttl_ms))
      continue;
    if (!prefix.empty() && key.rfind(prefix, 0) != 0) continue;
    ObjectSummary s;
    s.key = key;
    s.size = meta.size;
    s.ncopies = static_cast<uint32_t>(meta.copies.size());
    if (!meta.copies.empty() && !meta.copies[0].shards.empty())
      s.storage_class = meta.copies[0].shards[0].storage_class;
    out.push_back(std::move(s));
  }
  std::sort(out.begin(), out.end(),
            [](const ObjectSummary& a, const ObjectSummary& b) {
              return a.key < b.key;
            });
  return out;
}

uint64_t KeystoneService::remove_all_objects() {
  std::unique_lock lk(objects_mu_);
  uint64_t n = objects_.size();
  for (auto& [key, meta] : objects_) allocator_.free(key);
  objects_.clear();
  bump_placement_epoch_locked();
  bump_view();
  return n;
}

// ------------------------------------------------------------- batch ops

BatchPutStartResponse KeystoneService::batch_put_start(
    const std::vector<PutStartRequest>& reqs) {
  BatchPutStartResponse out;
  out.items.resize(reqs.size());
  if (!is_leader()) {
    for (auto& it : out.items)
      it.status = static_cast<int32_t>(ErrorCode::NOT_LEADER);
    out.view_version = view_version_.load();
    return out;
  }
  if (reqs.empty()) {
    out.view_version = view_version_.load();
    return out;
  }
  // uniform-config fast path: one objects lock + one allocator batch call
  bool uniform = true;
  for (size_t i = 1; i < reqs.size(); ++i) {
    const auto& a = reqs[i].config;
    const auto& b = reqs[0].config;
    if (a.replication != b.replication ||
        a.max_workers_per_copy != b.max_workers_per_copy ||
        a.preferred_class != b.preferred_class ||
        a.required_class != b.required_class || a.ttl_ms != b.ttl_ms) {
      uniform = false;
      break;
    }
  }
  if (!uniform) {
    for (size_t i = 0; i < reqs.size(); ++i) {
      auto res = put_start(reqs[i].key, reqs[i].size, reqs[i].config);
      if (res.ok()) out.items[i].copies = std::move(res.value().copies);
      else out.items[i].status = static_cast<int32_t>(res.code());
    }
    out.view_version = view_version_.load();
    return out;
  }

  // Lock discipline for scale-out (8 ranks hammer this concurrently): the
  // objects lock is held only for the map passes, never across allocation.
  std::vector<ObjectKey> keys;
  std::vector<uint64_t> sizes;
  std::vector<const ObjectKey*> replaced;  // freed OUTSIDE the objects lock
  keys.reserve(reqs.size());
  sizes.reserve(reqs.size());
  const uint64_t now = now_ms();
  {
    std::unique_lock lk(objects_mu_);
    // duplicate keys (vs existing objects) rejected up front
    for (size_t i = 0; i < reqs.size(); ++i) {
      auto it = objects_.find(reqs[i].key);
      if (it != objects_.end()) {
        if (!it->second.expired(now) && !reqs[i].config.replace) {
          out.items[i].status = static_cast<int32_t>(ErrorCode::OBJECT_EXISTS);
          keys.push_back({});  // hole keeps indices aligned
          sizes.push_back(0);
          continue;
        }
        ObjectMeta& m = it->second;
        if (reqs[i].config.replace && m.state == ObjectState::COMMITTED &&
            m.size == reqs[i].size &&
            m.copies.size() == std::max<uint32_t>(reqs[i].config.replication, 1)) {
          // same-size upsert: overwrite IN PLACE — no allocator free/alloc,
          // placements (and client placement caches) stay stable
          m.state = ObjectState::PENDING;
          m.checksum = 0;
          out.items[i].copies = m.copies;
          keys.push_back({});  // hole: nothing to allocate
          sizes.push_back(0);
          continue;
        }
        // upsert (size/shape changed) or expired: drop the meta here, free
        // the ranges in one allocator batch below (not per key under this
        // lock)
        objects_.erase(it);
        bump_placement_epoch_locked();
        mark_dirty_locked(reqs[i].key, true);
        replaced.push_back(&reqs[i].key);
      }
      keys.push_back(reqs[i].key);
      sizes.push_back(reqs[i].size);
    }
    if (!replaced.empty()) bump_view();
  }
  if (!replaced.empty()) allocator_.free_batch(replaced);
  // allocator has its own lock; two racing batches over the same fresh key
  // are serialized by the allocator's ledger (second gets OBJECT_EXISTS)
  auto placed = allocator_.allocate_batch(keys, sizes, reqs[0].config);
  // build metadata outside any lock
  std::vector<ObjectMeta> metas(reqs.size());
  for (size_t i = 0; i < reqs.size(); ++i) {
    if (out.items[i].status != 0 || keys[i].empty()) continue;
    if (placed[i].first != 0) {
      out.items[i].status = placed[i].first;
      continue;
    }
    ObjectMeta& meta = metas[i];
    meta.key = reqs[i].key;
    meta.size = reqs[i].size;
    meta.ttl_ms = reqs[i].config.ttl_ms ? reqs[i].config.ttl_ms
                                        : config_.object_ttl_default_ms;
    meta.created_ms = now;
    meta.last_access_ms = now;
    meta.replication = std::max<uint32_t>(reqs[i].config.replication, 1);
    meta.state = ObjectState::PENDING;
    meta.copies = placed[i].second;
  }
  {
    std::unique_lock lk(objects_mu_);
    objects_.reserve(objects_.size() + reqs.size());  // no mid-batch rehash
    for (size_t i = 0; i < reqs.size(); ++i) {
      if (out.items[i].status != 0 || keys[i].empty() || placed[i].first != 0)
        continue;
      auto [it, inserted] = objects_.try_emplace(reqs[i].key);
      if (!inserted) {
        // raced with a concurrent put of the same key: release our ranges
        out.items[i].status = static_cast<int32_t>(ErrorCode::OBJECT_EXISTS);
        lk.unlock();
        allocator_.free(reqs[i].key);
        lk.lock();
        continue;
      }
      it->second = std::move(metas[i]);
      out.items[i].copies = std::move(placed[i].second);
    }
    bump_view();
  }
  out.view_version = view_version_.load();
  return out;
}

std::vector<int32_t> KeystoneService::batch_put_complete(
    const std::vector<PutCompleteRequest>& reqs) {
  // one SHARED lock + one view bump for the whole batch: commit batches
  // from N ranks run concurrently (field mutations via atomic_ref — the
  // same concurrency argument as commit_token; structural changes hold the
  // lock exclusively and are therefore excluded)
  std::vector<int32_t> out(reqs.size(), 0);
  const uint64_t now = now_ms();
  bool any = false;
  {
    std::shared_lock lk(objects_mu_);
    for (size_t i = 0; i < reqs.size(); ++i) {
      auto it = objects_.find(reqs[i].key);
      if (it == objects_.end()) {
        out[i] = static_cast<int32_t>(ErrorCode::OBJECT_NOT_FOUND);
        continue;
      }
      ObjectMeta& m = it->second;
      const auto state = std::atomic_ref<ObjectState>(m.state)
                             .load(std::memory_order_acquire);
      if (state == ObjectState::COMMITTED) {
        // idempotent: a commit retried after a leader failover (the first
        // attempt applied + replicated, the reply was lost) carries the
        // same digest — report success, not INVALID_STATE
        out[i] = std::atomic_ref<uint64_t>(m.checksum)
                             .load(std::memory_order_relaxed) ==
                         reqs[i].checksum
                     ? 0
                     : static_cast<int32_t>(ErrorCode::INVALID_STATE);
        continue;
      }
      std::atomic_ref<uint64_t>(m.checksum)
          .store(reqs[i].checksum, std::memory_order_relaxed);
      apply_shard_digests_atomic(m, reqs[i].checksum, reqs[i].shard_digests);
      std::atomic_ref<uint64_t>(m.created_ms)
          .store(now, std::memory_order_relaxed);
      std::atomic_ref<uint64_t>(m.last_access_ms)
          .store(now, std::memory_order_relaxed);
      std::atomic_ref<ObjectState>(m.state)
          .store(ObjectState::COMMITTED, std::memory_order_release);
      mark_dirty_locked(reqs[i].key, false);
      any = true;
    }
    if (any) bump_view();
  }
  flush_dirty_now();
  return out;
}

// ------------------------------------------------- sessionful upserts

uint64_t KeystoneService::create_put_session(
    const std::vector<PutStartRequest>& reqs) {
  if (reqs.empty()) return 0;
  auto s = std::make_shared<PutSession>();
  s->metas.reserve(reqs.size());
  s->sizes.reserve(reqs.size());
  s->created_ms = now_ms();
  {
    std::shared_lock lk(objects_mu_);
    for (const auto& r : reqs) {
      auto it = objects_.find(r.key);
      // sessions cover single-SHARD copies (any replica count): the token
      // commit records ONE digest per object — every replica holds the
      // same bytes — and the client resolves one destination per copy
      if (it == objects_.end() || it->second.copies.empty())
        return 0;
      for (const auto& c : it->second.copies)
        if (c.shards.size() != 1) return 0;
      s->metas.push_back(&it->second);
      s->sizes.push_back(it->second.size);
    }
    s->epoch = placement_epoch_;
  }
  uint64_t token = next_session_token_.fetch_add(1);
  std::lock_guard<std::mutex> g(sessions_mu_);
  put_sessions_[token] = std::move(s);
  if (put_sessions_.size() > 1024) {
    // cap the table: evict the oldest session (tokens are monotonic)
    auto oldest = put_sessions_.begin();
    for (auto it = put_sessions_.begin(); it != put_sessions_.end(); ++it)
      if (it->first < oldest->first) oldest = it;
    put_sessions_.erase(oldest);
  }
  return token;
}

Result<void> KeystoneService::upsert_start_token(uint64_t token) {
  if (!is_leader())
    return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
  std::shared_ptr<PutSession> s;
  {
    std::lock_guard<std::mutex> g(sessions_mu_);
    auto it = put_sessions_.find(token);
    if (it == put_sessions_.end())
      return Error{ErrorCode::SESSION_STALE, "unknown put session"};
    s = it->second;
  }
  // SHARED lock: session steps from N ranks run concurrently (the warm
  // plane would otherwise convoy on one mutex at 8 ranks × 2 lanes).
  // Removal/GC/eviction hold the lock exclusively, so the session's meta
  // pointers cannot be invalidated inside this section; mutated fields go
  // through atomic_ref because another session on the SAME keys may write
  // them concurrently (same-key concurrent upserts are last-writer-wins,
  // consistent with in-place data overwrites being racy by design).
  std::shared_lock lk(objects_mu_);
  if (s->epoch != placement_epoch_)
    return Error{ErrorCode::SESSION_STALE, "placements changed"};
  // validate ALL before flipping ANY (all-or-nothing)
  for (size_t i = 0; i < s->metas.size(); ++i) {
    const ObjectMeta* m = s->metas[i];
    // PENDING allowed: a client that died between start and commit may
    // retry the same session
    if (m->size != s->sizes[i])
      return Error{ErrorCode::SESSION_STALE, "object shape changed"};
  }
  // PENDING pins the placements: tiering/eviction/repair/scrub only touch
  // COMMITTED objects, so the client's one-sided writes land in ranges that
  // cannot move underneath them
  for (auto* m : s->metas)
    std::atomic_ref<ObjectState>(m->state)
        .store(ObjectState::PENDING, std::memory_order_relaxed);
  return {};
}

Result<void> KeystoneService::commit_token(
    uint64_t token, const std::vector<uint64_t>& digests, bool release) {
  if (!is_leader())
    return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
  std::shared_ptr<PutSession> s;
  {
    std::lock_guard<std::mutex> g(sessions_mu_);
    auto it = put_sessions_.find(token);
    if (it == put_sessions_.end())
      return Error{ErrorCode::SESSION_STALE, "unknown put session"};
    s = it->second;
  }
  if (digests.size() != s->metas.size())
    return Error{ErrorCode::INVALID_ARGUMENT, "digest count mismatch"};
  const uint64_t now = now_ms();
  // SHARED lock — see upsert_start_token for the concurrency argument
  std::shared_lock lk(objects_mu_);
  if (s->epoch != placement_epoch_)
    return Error{ErrorCode::SESSION_STALE, "placements changed"};
  for (size_t i = 0; i < s->metas.size(); ++i)
    if (s->metas[i]->size != s->sizes[i])
      return Error{ErrorCode::SESSION_STALE, "object shape changed"};
  const bool persist = config_.persist_objects;
  const size_t n = s->metas.size();
  for (size_t i = 0; i < n; ++i) {
    // metas are scattered map nodes: prefetch ahead — this loop is in the
    // hot warm-step path (one commit per session put step)
    if (i + 8 < n) __builtin_prefetch(s->metas[i + 8], 1, 1);
    ObjectMeta* m = s->metas[i];
    std::atomic_ref<uint64_t>(m->checksum)
        .store(digests[i], std::memory_order_relaxed);
    // single-shard copies by construction; every replica gets the digest
    for (auto& c : m->copies)
      std::atomic_ref<uint64_t>(c.shards[0].digest)
          .store(digests[i], std::memory_order_relaxed);
    std::atomic_ref<uint64_t>(m->created_ms)
        .store(now, std::memory_order_relaxed);  // TTL restarts at commit
    std::atomic_ref<uint64_t>(m->last_access_ms)
        .store(now, std::memory_order_relaxed);
    std::atomic_ref<ObjectState>(m->state)
        .store(ObjectState::COMMITTED, std::memory_order_release);
    if (persist) mark_dirty_locked(m->key, false);
  }
  bump_view();
  ctr_token_commits_.fetch_add(1);
  lk.unlock();
  if (release) {
    std::lock_guard<std::mutex> g(sessions_mu_);
    put_sessions_.erase(token);
  }
  flush_dirty_now();
  return {};
}

std::vector<int32_t> KeystoneService::batch_put_cancel(
    const std::vector<ObjectKey>& keys) {
  std::vector<int32_t> out;
  out.reserve(keys.size());
  for (const auto& k : keys)
    out.push_back(static_cast<int32_t>(put_cancel(k).code()));
  return out;
}

BatchGetWorkersResponse KeystoneService::batch_get_workers(
    const std::vector<ObjectKey>& keys) {
  // SHARED lock: get batches from N ranks run concurrently with each other
  // and with session commits. Fields a concurrent commit mutates (state,
  // checksum, shard digest, access times) are read through atomic_ref; the
  // placement STRUCTURE (strings, offsets, vector shapes) only changes
  // under the exclusive lock, so plain reads of those are safe. Expired
  // objects are collected and removed in a rare second exclusive pass.
  BatchGetWorkersResponse out;
  out.items.resize(keys.size());
  const uint64_t now = now_ms();
  std::vector<size_t> expired_idx;
  {
    std::shared_lock lk(objects_mu_);
    for (size_t i = 0; i < keys.size(); ++i) {
      auto& item = out.items[i];
      auto it = objects_.find(keys[i]);
      if (it == objects_.end()) {
        item.status = static_cast<int32_t>(ErrorCode::OBJECT_NOT_FOUND);
        continue;
      }
      auto& meta = it->second;
      const auto state = std::atomic_ref<ObjectState>(meta.state)
                             .load(std::memory_order_acquire);
      if (state != ObjectState::COMMITTED) {
        item.status = static_cast<int32_t>(ErrorCode::OBJECT_NOT_COMMITTED);
        continue;
      }
      const uint64_t created = std::atomic_ref<uint64_t>(meta.created_ms)
                                   .load(std::memory_order_relaxed);
      if (meta.ttl_ms > 0 && now > created + meta.ttl_ms) {
        expired_idx.push_back(i);
        item.status = static_cast<int32_t>(ErrorCode::OBJECT_EXPIRED);
        continue;
      }
      std::atomic_ref<uint64_t>(meta.last_access_ms)
          .store(now, std::memory_order_relaxed);
      std::atomic_ref<uint32_t>(meta.access_count)
          .fetch_add(1, std::memory_order_relaxed);
      item.status = 0;
      item.info.size = meta.size;
      item.info.checksum = std::atomic_ref<uint64_t>(meta.checksum)
                               .load(std::memory_order_relaxed);
      item.info.copies.reserve(meta.copies.size());
      for (auto& c : meta.copies) {
        CopyPlacement cp;
        cp.copy_index = c.copy_index;
        cp.shards.reserve(c.shards.size());
        for (auto& sh : c.shards) {
          ShardPlacement sp;
          sp.pool_id = sh.pool_id;
          sp.worker_id = sh.worker_id;
          sp.storage_class = sh.storage_class;
          sp.offset = sh.offset;
          sp.length = sh.length;
          sp.digest = std::atomic_ref<uint64_t>(sh.digest)
                          .load(std::memory_order_relaxed);
          sp.access = sh.access;
          cp.shards.push_back(std::move(sp));
        }
        item.info.copies.push_back(std::move(cp));
      }
    }
  }
  if (!expired_idx.empty()) {
    std::unique_lock lk(objects_mu_);
    for (size_t i : expired_idx) {
      auto it = objects_.find(keys[i]);
      if (it != objects_.end() && it->second.expired(now))
        remove_object_locked(keys[i]);
    }
  }
  return out;
}

std::vector<uint8_t> KeystoneService::batch_object_exists(
    const std::vector<ObjectKey>& keys) {
  std::vector<uint8_t> out;
  out.reserve(keys.size());
  for (const auto& k : keys) out.push_back(object_exists(k) ? 1 : 0);
  return out;
}

std::vector<int32_t> KeystoneService::batch_remove(
    const std::vector<ObjectKey>& keys) {
  std::vector<int32_t> out(keys.size(), 0);
  std::vector<const ObjectKey*> to_free;
  to_free.reserve(keys.size());
  {
    std::unique_lock lk(objects_mu_);
    for (size_t i = 0; i < keys.size(); ++i) {
      auto it = objects_.find(keys[i]);
      if (it == objects_.end()) {
        out[i] = static_cast<int32_t>(ErrorCode::OBJECT_NOT_FOUND);
        continue;
      }
      objects_.erase(it);
      bump_placement_epoch_locked();
      mark_dirty_locked(keys[i], true);
      to_free.push_back(&keys[i]);
    }
    bump_view();
  }
  // range frees take only the allocator's own lock — once for the batch
  allocator_.free_batch(to_free);
  flush_dirty_now();  // deletions durable before the reply (retry-safe)
  return out;
}

// ------------------------------------------------------------ cluster view

std::vector<WorkerInfo> KeystoneService::get_workers_info() {
  std::shared_lock lk(workers_mu_);
  std::vector<WorkerInfo> out;
  out.reserve(workers_.size());
  for (const auto& [id, w] : workers_) out.push_back(w);
  return out;
}

std::vector<MemoryPool> KeystoneService::get_memory_pools() {
  return allocator_.pools();
}

Result<void> KeystoneService::remove_worker(const WorkerId& id) {
  {
    std::shared_lock lk(workers_mu_);
    if (!workers_.count(id)) return Error{ErrorCode::KEY_NOT_FOUND, id};
  }
  // delete the coordination keys; the watchers do the rest (same path as a
  // TTL death). Reference left this unimplemented (keystone_service.cpp:125).
  coord_->del(prefix() + "/heartbeat/" + id);
  coord_->del(prefix() + "/workers/" + id);
  auto pools = coord_->get_prefix(prefix() + "/memory_pools/" + id + "/");
  if (pools.ok())
    for (const auto& kv : pools.value()) coord_->del(kv.key);
  cleanup_dead_worker(id);
  return {};
}

ClusterStats KeystoneService::get_cluster_stats() {
  ClusterStats s;
  auto as = allocator_.stats();
  s.total_capacity = as.total_capacity;
  s.total_used = as.total_used;
  s.num_pools = as.num_pools;
  {
    std::shared_lock lk(objects_mu_);
    s.num_objects = objects_.size();
  }
  {
    std::shared_lock lk(workers_mu_);
    s.num_workers = workers_.size();
  }
  s.view_version = view_version_.load();
  return s;
}

void KeystoneService::register_pool(const MemoryPool& pool) {
  allocator_.upsert_pool(pool);
  bump_view();
}

void KeystoneService::register_worker(const WorkerInfo& info) {
  std::unique_lock lk(workers_mu_);
  workers_[info.worker_id] = info;
  bump_view();
}

// ---------------------------------------------------------- maintenance

void KeystoneService::run_gc_once() {
  uint64_t now = now_ms();
  std::unique_lock lk(objects_mu_);
  std::vector<ObjectKey> dead;
  for (const auto& [key, meta] : objects_) {
    if (meta.expired(now)) dead.push_back(key);
    // abandoned PENDING puts: reclaim after 10 minutes
    else if (meta.state == ObjectState::PENDING &&
             now > meta.created_ms + 600000)
      dead.push_back(key);
  }
  for (const auto& k : dead) remove_object_locked(k);
  if (!dead.empty()) {
    ctr_gc_.fetch_add(dead.size());
    BB_LOG(DEBUG) << "gc reclaimed " << dead.size() << " objects";
  }
}

void KeystoneService::run_eviction_once() {
  auto as = allocator_.stats();
  if (as.total_capacity == 0) return;
  double fill = static_cast<double>(as.total_used) / as.total_capacity;
  if (fill < config_.eviction_high_watermark) return;

  std::unique_lock lk(objects_mu_);
  // age-based: evict least-recently-accessed committed objects
  std::vector<std::pair<uint64_t, ObjectKey>> cand;
  for (const auto& [key, meta] : objects_)
    if (meta.state == ObjectState::COMMITTED)
      cand.emplace_back(meta.last_access_ms, key);
  std::sort(cand.begin(), cand.end());
  size_t target = static_cast<size_t>(cand.size() * config_.eviction_ratio) + 1;
  size_t evicted = 0;
  for (const auto& [ts, key] : cand) {
    if (evicted >= target) break;
    remove_object_locked(key);
    ++evicted;
  }
  ctr_evictions_.fetch_add(evicted);
  BB_LOG(INFO) << "eviction: fill " << fill << " → evicted " << evicted
               << " objects";
}

void KeystoneService::gc_loop() {
  while (running_) {
    {
      std::unique_lock<std::mutex> lk(cv_mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(config_.gc_interval_ms),
                   [this] { return !running_.load(); });
    }
    if (!running_) break;
    if (!is_leader()) continue;  // standby: the leader owns maintenance
    run_gc_once();
    run_repair_once();
    if (config_.enable_tiering) run_tiering_once();
    if (config_.compact_fragmentation_threshold > 0) run_compaction_once();
    if (config_.scrub_interval_ms > 0) run_scrub_once();
    run_eviction_once();
  }
}

void KeystoneService::set_advertised_endpoint(const std::string& ep) {
  // adv_mu_ held ACROSS the registry write: otherwise the keepalive thread
  // can read the empty advertised endpoint, lose the race, and overwrite
  // the registry with the configured (possibly port-0) listen address
  std::lock_guard<std::mutex> g(adv_mu_);
  advertised_endpoint_ = ep;
  coord_->put("/blackbird/services/blackbird-keystone/" + instance_id_, ep,
              config_.worker_ttl_ms * 6);
}

void KeystoneService::keepalive_loop() {
  const std::string key = "/blackbird/services/blackbird-keystone/" + instance_id_;
  while (running_) {
    {
      std::lock_guard<std::mutex> g(adv_mu_);
      const std::string& ep = advertised_endpoint_.empty()
                                  ? config_.listen_address
                                  : advertised_endpoint_;
      coord_->put(key, ep, config_.worker_ttl_ms * 6);
    }
    std::unique_lock<std::mutex> lk(cv_mu_);
    cv_.wait_for(lk, std::chrono::milliseconds(config_.worker_ttl_ms * 3),
                 [this] { return !running_.load(); });
  }
}


// ---------------------------------------------------------- tier migration

rpc::RpcClient* KeystoneService::data_client(const std::string& endpoint) {
  std::lock_guard<std::mutex> g(data_clients_mu_);
  auto& pool = data_clients_[endpoint];
  // round-robin over up to kDataConns live connections (grown on demand);
  // a dead connection is replaced in its slot
  if (pool.conns.size() < kDataConns) {
    auto c = std::make_unique<rpc::RpcClient>();
    if (!c->connect(endpoint).ok())
      return pool.conns.empty() ? nullptr : pool.conns[0].get();
    pool.conns.push_back(std::move(c));
    return pool.conns.back().get();
  }
  pool.cursor = (pool.cursor + 1) % pool.conns.size();
  auto& slot = pool.conns[pool.cursor];
  if (!slot->connected()) {
    auto c = std::make_unique<rpc::RpcClient>();
    if (!c->connect(endpoint).ok()) return nullptr;
    slot = std::move(c);
  }
  return slot.get();
}

namespace {
struct PullReq {
  std::string dst_pool;
  uint64_t dst_offset = 0;
  uint64_t total_len = 0;
  std::vector<ShardPlacement> srcs;
  BB_FIELDS(dst_pool, dst_offset, total_len, srcs)
};

// Sub-ranges of ordered shards `srcs` (covering [0, total)) that cover the
// object range [a, b).
std::vector<ShardPlacement> slice_shards(const std::vector<ShardPlacement>& srcs,
                                         uint64_t a, uint64_t b) {
  std::vector<ShardPlacement> out;
  uint64_t off = 0;
  for (const auto& s : srcs) {
    uint64_t s_begin = off, s_end = off + s.length;
    off = s_end;
    uint64_t lo = std::max(a, s_begin), hi = std::min(b, s_end);
    if (lo >= hi) continue;
    ShardPlacement part = s;
    part.offset = s.offset + (lo - s_begin);
    part.length = hi - lo;
    out.push_back(std::move(part));
  }
  return out;
}
}  // namespace

Result<void> KeystoneService::migrate_object(const ObjectKey& key,
                                             StorageClass target) {
  // snapshot the object (no lock held during the transfer). EXCLUSIVE so
  // the full-struct copy cannot race shared-lock session commits.
  ObjectMeta snap;
  {
    std::unique_lock lk(objects_mu_);
    auto it = objects_.find(key);
    if (it == objects_.end()) return Error{ErrorCode::OBJECT_NOT_FOUND, key};
    if (it->second.state != ObjectState::COMMITTED)
      return Error{ErrorCode::OBJECT_NOT_COMMITTED, key};
    if (it->second.copies.empty())
      return Error{ErrorCode::INVALID_STATE, "object has no copies"};
    snap = it->second;
  }

  // allocate the destination placement under a temp ledger key — same copy
  // count as the source; prefer one shard per copy, fall back to striping
  // when the target tier is too fragmented for a contiguous range
  const std::string tmp_key = key + "\x01mig";
  PlacementConfig mcfg;
  mcfg.replication = static_cast<uint32_t>(snap.copies.size());
  mcfg.max_workers_per_copy = 1;
  mcfg.required_class = target;
  auto placed = allocator_.allocate(tmp_key, snap.size, mcfg);
  if (!placed.ok()) {
    mcfg.max_workers_per_copy = 4;
    placed = allocator_.allocate(tmp_key, snap.size, mcfg);
    if (!placed.ok()) return placed.error();
  }

  // each destination shard pulls its slice of the matching source copy
  // (src shards enriched with access info so the puller can reach remote
  // pools)
  for (size_t ci = 0; ci < placed.value().size(); ++ci) {
    const auto& src_shards = snap.copies[ci % snap.copies.size()].shards;
    uint64_t off = 0;
    for (const auto& dst_shard : placed.value()[ci].shards) {
      PullReq req;
      req.dst_pool = dst_shard.pool_id;
      req.dst_offset = dst_shard.offset;
      req.total_len = dst_shard.length;
      req.srcs = slice_shards(src_shards, off, off + dst_shard.length);
      for (auto& sh : req.srcs) {
        auto a = allocator_.pool_access(sh.pool_id);
        if (a.ok()) sh.access = std::move(a.value());
      }
      off += dst_shard.length;
      auto dst_access = allocator_.pool_access(dst_shard.pool_id);
      if (!dst_access.ok()) {
        allocator_.free(tmp_key);
        return dst_access.error();
      }
      auto* dc = data_client(dst_access.value().endpoint);
      Result<std::string> pulled =
          dc ? dc->call_raw(rpc::methods::DATA_PULL, serde::to_bytes(req),
                            120000)
             : Result<std::string>(Error{ErrorCode::CONNECT_FAILED,
                                         dst_access.value().endpoint});
      if (!pulled.ok()) {
        allocator_.free(tmp_key);
        return pulled.error();
      }
    }
  }

  // commit the move: re-validate, swap placement, release the old ranges
  {
    std::unique_lock lk(objects_mu_);
    auto it = objects_.find(key);
    bool unchanged = it != objects_.end() &&
                     it->second.state == ObjectState::COMMITTED &&
                     it->second.copies.size() == snap.copies.size();
    for (size_t ci = 0; unchanged && ci < snap.copies.size(); ++ci)
      unchanged = it->second.copies[ci].shards == snap.copies[ci].shards;
    if (!unchanged) {
      lk.unlock();
      allocator_.free(tmp_key);
      return Error{ErrorCode::INVALID_STATE, "object changed during migration"};
    }
    allocator_.free(key);
    auto rn = allocator_.rename(tmp_key, key);
    if (!rn.ok()) {
      // old ranges already freed; keep the new placement under tmp is wrong —
      // this cannot happen (key was just freed), but guard anyway
      BB_LOG(ERROR) << "migration rename failed: " << rn.message();
    }
    it->second.copies = std::move(placed.value());
    if (it->second.copies.size() == 1 &&
        it->second.copies[0].shards.size() == 1)
      it->second.copies[0].shards[0].digest = it->second.checksum;
    it->second.access_count = 0;
    bump_placement_epoch_locked();
    mark_dirty_locked(key, false);
    bump_view();
  }
  ctr_migrations_.fetch_add(1);
  BB_LOG(INFO) << "migrated " << key << " → " << to_string(target);
  return {};
}

void KeystoneService::run_tiering_once() {
  // per-tier fill from the allocator's pool view
  struct Agg {
    uint64_t cap = 0, used = 0;
  };
  std::map<int, Agg> tiers;  // tier_rank → agg
  std::map<int, StorageClass> rank_class;
  for (const auto& p : allocator_.pools()) {
    int r = tier_rank(p.storage_class);
    tiers[r].cap += p.size;
    tiers[r].used += p.used;
    rank_class[r] = p.storage_class;
  }
  if (tiers.empty()) return;

  uint32_t moves_left = config_.tier_max_moves_per_cycle;

  // ---- demotion: fastest overfull tier → next tier with room ----
  for (auto it = tiers.begin(); it != tiers.end() && moves_left > 0; ++it) {
    auto [rank, agg] = *it;
    if (agg.cap == 0) continue;
    double fill = static_cast<double>(agg.used) / agg.cap;
    if (fill <= config_.tier_high_watermark) continue;
    // find the next tier down with capacity headroom
    StorageClass target{};
    bool found = false;
    for (auto jt = std::next(it); jt != tiers.end(); ++jt) {
      double jf = jt->second.cap
                      ? static_cast<double>(jt->second.used) / jt->second.cap
                      : 1.0;
      if (jf < config_.tier_high_watermark) {
        target = rank_class[jt->first];
        found = true;
        break;
      }
    }
    if (!found) continue;

    // LRU single-copy committed objects living in this tier
    std::vector<std::pair<uint64_t, ObjectKey>> cands;
    {
      std::shared_lock lk(objects_mu_);
      for (const auto& [key, meta] : objects_) {
        if (rload(meta.state) != ObjectState::COMMITTED ||
            meta.copies.size() != 1)
          continue;
        bool in_tier = !meta.copies[0].shards.empty();
        for (const auto& sh : meta.copies[0].shards)
          if (tier_rank(sh.storage_class) != rank) in_tier = false;
        if (in_tier) cands.emplace_back(rload(meta.last_access_ms), key);
      }
    }
    std::sort(cands.begin(), cands.end());
    uint64_t bytes_over = agg.used - static_cast<uint64_t>(
                                         agg.cap * config_.tier_high_watermark);
    // pick the batch up-front, then migrate with a small worker fan-out
    // (each migration is an independent pull; the data plane overlaps)
    std::vector<ObjectKey> batch;
    uint64_t planned = 0;
    for (const auto& [ts, key] : cands) {
      if (batch.size() >= moves_left || planned >= bytes_over) break;
      uint64_t sz = 0;
      {
        std::shared_lock lk(objects_mu_);
        auto oit = objects_.find(key);
        if (oit == objects_.end()) continue;
        sz = oit->second.size;
      }
      batch.push_back(key);
      planned += sz;
    }
    if (!batch.empty()) {
      std::atomic<size_t> next_idx{0};
      std::atomic<uint32_t> done{0};
      const int nthreads = std::min<int>(4, static_cast<int>(batch.size()));
      std::vector<std::thread> ts2;
      for (int t = 0; t < nthreads; ++t)
        ts2.emplace_back([&] {
          for (size_t bi = next_idx.fetch_add(1); bi < batch.size();
               bi = next_idx.fetch_add(1))
            if (migrate_object(batch[bi], target).ok()) done.fetch_add(1);
        });
      for (auto& t2 : ts2) t2.join();
      moves_left -= std::min(moves_left, done.load());
    }
  }

  // ---- promotion: hot objects move toward the fastest tier with room ----
  if (config_.promote_hot_threshold > 0 && moves_left > 0) {
    auto fastest = tiers.begin();
    if (fastest->second.cap > 0) {
      double fill =
          static_cast<double>(fastest->second.used) / fastest->second.cap;
      if (fill < config_.tier_high_watermark * 0.9) {
        StorageClass target = rank_class[fastest->first];
        std::vector<std::pair<uint32_t, ObjectKey>> hot;
        {
          std::shared_lock lk(objects_mu_);
          for (const auto& [key, meta] : objects_) {
            if (rload(meta.state) != ObjectState::COMMITTED ||
                meta.copies.size() != 1 || meta.copies[0].shards.empty())
              continue;
            const uint32_t ac = rload(meta.access_count);
            if (ac < config_.promote_hot_threshold) continue;
            if (tier_rank(meta.copies[0].shards[0].storage_class) >
                fastest->first)
              hot.emplace_back(ac, key);
          }
        }
        std::sort(hot.rbegin(), hot.rend());  // hottest first
        for (const auto& [cnt, key] : hot) {
          if (moves_left == 0) break;
          if (migrate_object(key, target).ok()) --moves_left;
        }
      }
    }
  }

  // decay access counts so "hot" means hot recently (only meaningful when
  // promotion is on — skip the full-map pass otherwise)
  if (config_.promote_hot_threshold > 0) {
    std::unique_lock lk(objects_mu_);
    for (auto& [key, meta] : objects_) meta.access_count /= 2;
  }
}


// ---------------------------------------------------------- failure repair

Result<void> KeystoneService::repair_object(const ObjectKey& key) {
  ObjectMeta snap;  // exclusive: full-struct copy (see migrate_object)
  {
    std::unique_lock lk(objects_mu_);
    auto it = objects_.find(key);
    if (it == objects_.end()) return Error{ErrorCode::OBJECT_NOT_FOUND, key};
    if (it->second.state != ObjectState::COMMITTED)
      return Error{ErrorCode::OBJECT_NOT_COMMITTED, key};
    snap = it->second;
  }
  if (snap.copies.empty())
    return Error{ErrorCode::NO_PLACEMENT, key};
  if (snap.copies.size() >= snap.replication) return {};

  // allocate one extra copy on workers not already holding the object
  std::vector<WorkerId> holders;
  uint32_t max_idx = 0;
  for (const auto& c : snap.copies) {
    max_idx = std::max(max_idx, c.copy_index);
    for (const auto& sh : c.shards) holders.push_back(sh.worker_id);
  }
  PlacementConfig rcfg;
  rcfg.replication = 1;
  rcfg.max_workers_per_copy = 1;
  const std::string tmp_key = key + "\x01rep";
  auto placed = allocator_.allocate_extra_copy(tmp_key, snap.size, rcfg,
                                               max_idx + 1, holders);
  if (!placed.ok()) return placed.error();

  // pull the bytes from the surviving copy into each new shard
  const auto& src_shards = snap.copies[0].shards;
  uint64_t off = 0;
  bool ok = true;
  Error last{ErrorCode::TRANSFER_FAILED, "repair pull"};
  for (const auto& dst : placed.value().shards) {
    PullReq req;
    req.dst_pool = dst.pool_id;
    req.dst_offset = dst.offset;
    req.total_len = dst.length;
    req.srcs = slice_shards(src_shards, off, off + dst.length);
    for (auto& sh : req.srcs) {
      auto a = allocator_.pool_access(sh.pool_id);
      if (a.ok()) sh.access = std::move(a.value());
    }
    off += dst.length;
    auto dst_access = allocator_.pool_access(dst.pool_id);
    if (!dst_access.ok()) {
      ok = false;
      last = dst_access.error();
      break;
    }
    auto* dc = data_client(dst_access.value().endpoint);
    if (!dc) {
      ok = false;
      last = Error{ErrorCode::CONNECT_FAILED, dst_access.value().endpoint};
      break;
    }
    auto r = dc->call_raw(rpc::methods::DATA_PULL, serde::to_bytes(req), 120000);
    if (!r.ok()) {
      ok = false;
      last = r.error();
      break;
    }
  }
  if (!ok) {
    allocator_.free(tmp_key);
    return last;
  }

  {
    std::unique_lock lk(objects_mu_);
    auto it = objects_.find(key);
    if (it == objects_.end() || it->second.state != ObjectState::COMMITTED ||
        it->second.copies.size() != snap.copies.size()) {
      lk.unlock();
      allocator_.free(tmp_key);
      return Error{ErrorCode::INVALID_STATE, "object changed during repair"};
    }
    auto mr = allocator_.merge_into(tmp_key, key);
    if (!mr.ok()) {
      lk.unlock();
      allocator_.free(tmp_key);
      return mr.error();
    }
    it->second.copies.push_back(std::move(placed.value()));
    bump_placement_epoch_locked();
    mark_dirty_locked(key, false);
    bump_view();
  }
  ctr_repairs_.fetch_add(1);
  BB_LOG(INFO) << "repaired " << key << " (copies "
               << snap.copies.size() << " → " << snap.copies.size() + 1 << ")";
  return {};
}

void KeystoneService::run_repair_once() {
  std::vector<ObjectKey> degraded;
  {
    std::shared_lock lk(objects_mu_);
    for (const auto& [key, meta] : objects_)
      if (rload(meta.state) == ObjectState::COMMITTED &&
          !meta.copies.empty() && meta.copies.size() < meta.replication)
        degraded.push_back(key);
  }
  if (degraded.size() > config_.repair_max_per_cycle)
    degraded.resize(config_.repair_max_per_cycle);
  // fan the re-replication pulls out over a few threads — after a worker
  // death the backlog is bandwidth-bound on the surviving workers, not on
  // keystone
  std::atomic<size_t> next{0};
  auto drain = [&] {
    for (size_t i = next.fetch_add(1); i < degraded.size();
         i = next.fetch_add(1)) {
      auto r = repair_object(degraded[i]);
      if (!r.ok() && r.code() != ErrorCode::NO_SPACE)
        BB_LOG(WARN) << "repair of " << degraded[i]
                     << " failed: " << r.message();
    }
  };
  size_t nthreads = std::min<size_t>(4, degraded.size());
  if (nthreads <= 1) {
    drain();
  } else {
    std::vector<std::thread> ts;
    for (size_t t = 0; t < nthreads; ++t) ts.emplace_back(drain);
    for (auto& t : ts) t.join();
  }
}


// ------------------------------------------------------- object persistence

void KeystoneService::mark_dirty_locked(const ObjectKey& key, bool removed) {
  if (!config_.persist_objects) return;
  std::lock_guard<std::mutex> g(dirty_mu_);
  dirty_[key] = removed;
}

// Synchronous durability for commit/remove acknowledgements: drain the dirty
// set NOW in one coordination round trip (put_many) so an acked mutation
// survives an immediate leader crash — the 100 ms persist_loop stays as the
// backstop for maintenance-path mutations (tiering, repair, scrub).
// Must be called WITHOUT objects_mu_ held (serializes dirty→blob under a
// shared lock, same order as persist_loop).
void KeystoneService::flush_dirty_now() {
  if (!config_.persist_objects || !coord_) return;
  std::map<ObjectKey, bool> batch;
  {
    std::lock_guard<std::mutex> g(dirty_mu_);
    if (dirty_.empty()) return;
    batch.swap(dirty_);
  }
  const std::string obj_prefix = prefix() + "/objects/";
  std::vector<coord::KV> puts;
  std::vector<std::string> dels;
  {
    // EXCLUSIVE: serde reads every meta field; a concurrent shared-lock
    // commit of the same key must not interleave with the serialization
    std::unique_lock lk(objects_mu_);
    for (const auto& [key, removed] : batch) {
      if (removed) {
        dels.push_back(obj_prefix + key);
        continue;
      }
      auto it = objects_.find(key);
      if (it == objects_.end()) continue;
      puts.push_back(coord::KV{obj_prefix + key, serde::to_bytes(it->second)});
    }
  }
  auto r = coord_->put_many(puts, dels);
  if (!r.ok()) {
    // coordination hiccup: put the work back for the async loop to retry
    BB_LOG(WARN) << "sync persist failed (" << r.error().message
                 << "); re-queued for async flush";
    std::lock_guard<std::mutex> g(dirty_mu_);
    for (auto& [key, removed] : batch) dirty_.emplace(key, removed);
  }
}

void KeystoneService::persist_loop() {
  const std::string obj_prefix = prefix() + "/objects/";
  while (running_) {
    {
      std::unique_lock<std::mutex> lk(cv_mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(100),
                   [this] { return !running_.load(); });
    }
    std::map<ObjectKey, bool> batch;
    {
      std::lock_guard<std::mutex> g(dirty_mu_);
      batch.swap(dirty_);
    }
    for (const auto& [key, removed] : batch) {
      if (removed) {
        coord_->del(obj_prefix + key);
        continue;
      }
      std::string blob;
      {
        std::unique_lock lk(objects_mu_);  // see flush_dirty_now
        auto it = objects_.find(key);
        if (it == objects_.end()) continue;
        blob = serde::to_bytes(it->second);
      }
      coord_->put(obj_prefix + key, blob, 0);
    }
  }
  // final flush on shutdown
  std::map<ObjectKey, bool> batch;
  {
    std::lock_guard<std::mutex> g(dirty_mu_);
    batch.swap(dirty_);
  }
  for (const auto& [key, removed] : batch) {
    if (removed) {
      coord_->del(obj_prefix + key);
    } else {
      std::unique_lock lk(objects_mu_);  // see flush_dirty_now
      auto it = objects_.find(key);
      if (it != objects_.end())
        coord_->put(obj_prefix + key, serde::to_bytes(it->second), 0);
    }
  }
}


// ------------------------------------------------------------- compaction

Result<uint32_t> KeystoneService::compact_pool(const PoolId& pool_id,
                                               uint32_t max_moves) {
  // find the pool and its class
  StorageClass cls{};
  bool found = false;
  for (const auto& p : allocator_.pools()) {
    if (p.pool_id == pool_id) {
      cls = p.storage_class;
      found = true;
      break;
    }
  }
  if (!found) return Error{ErrorCode::POOL_NOT_FOUND, pool_id};

  // single-copy objects with a shard in this pool, largest offset first —
  // re-allocating them walks data toward the low end and coalesces holes
  std::vector<std::pair<uint64_t, ObjectKey>> cands;
  {
    std::shared_lock lk(objects_mu_);
    for (const auto& [key, meta] : objects_) {
      if (rload(meta.state) != ObjectState::COMMITTED ||
          meta.copies.size() != 1)
        continue;
      for (const auto& sh : meta.copies[0].shards)
        if (sh.pool_id == pool_id) {
          cands.emplace_back(sh.offset, key);
          break;
        }
    }
  }
  std::sort(cands.rbegin(), cands.rend());
  uint32_t moved = 0;
  for (const auto& [off, key] : cands) {
    if (moved >= max_moves) break;
    // migrate within the same tier: the allocator's best-fit will prefer
    // the coalesced low-offset holes
    auto r = migrate_object(key, cls);
    if (r.ok()) ++moved;
  }
  return moved;
}

namespace {
struct ScrubChecksumReq {
  std::string pool_id;
  uint64_t offset = 0;
  uint64_t length = 0;
  BB_FIELDS(pool_id, offset, length)
};
struct ScrubU64Msg {
  uint64_t v = 0;
  BB_FIELDS(v)
};
}  // namespace

uint32_t KeystoneService::run_scrub_once(uint32_t max_objects) {
  if (max_objects == 0) max_objects = config_.scrub_batch;
  const uint64_t now = now_ms();
  struct Cand {
    ObjectKey key;
    uint64_t checksum;
    // every copy, striped or not: shard digests recorded at put time make
    // each shard independently verifiable (the round-1 scrubber silently
    // skipped striped copies — exactly the objects with the most failure
    // surface)
    std::vector<std::vector<ShardPlacement>> copies;
  };
  std::vector<Cand> cands;
  {
    std::unique_lock lk(objects_mu_);
    for (auto& [key, meta] : objects_) {
      if (cands.size() >= max_objects) break;
      if (meta.state != ObjectState::COMMITTED || meta.checksum == 0)
        continue;
      if (meta.last_scrub_ms != 0 &&
          now < meta.last_scrub_ms + config_.scrub_interval_ms)
        continue;
      if (meta.copies.empty()) continue;
      meta.last_scrub_ms = now;
      Cand cd;
      cd.key = key;
      cd.checksum = meta.checksum;
      for (const auto& c : meta.copies) cd.copies.push_back(c.shards);
      cands.push_back(std::move(cd));
    }
  }

  uint32_t quarantined = 0;
  for (const auto& cd : cands) {
    for (const auto& shards : cd.copies) {
      // verify every shard of the copy; ONE bad shard quarantines the copy
      bool copy_bad = false;
      bool verified_any = false;
      const ShardPlacement* bad_sh = nullptr;
      uint64_t bad_want = 0, bad_got = 0;
      for (const auto& sh : shards) {
        const uint64_t want =
            sh.digest != 0 ? sh.digest
                           : (shards.size() == 1 ? cd.checksum : 0);
        if (want == 0) continue;  // digest never recorded (e.g. repair copy)
        auto access = allocator_.pool_access(sh.pool_id);
        if (!access.ok()) continue;  // pool gone: dead-worker cleanup owns it
        auto* dc = data_client(access.value().endpoint);
        if (!dc) continue;
        ScrubChecksumReq req{sh.pool_id, sh.offset, sh.length};
        auto resp = dc->call_raw(rpc::methods::DATA_CHECKSUM,
                                 serde::to_bytes(req), 60000);
        if (!resp.ok()) continue;  // transient worker trouble: retry next pass
        ScrubU64Msg got;
        if (!serde::from_bytes(resp.value(), got)) continue;
        verified_any = true;
        if (got.v != want) {
          copy_bad = true;
          bad_sh = &sh;
          bad_want = want;
          bad_got = got.v;
          break;
        }
      }
      if (!verified_any || !copy_bad) continue;

      // corrupt copy: quarantine it (re-validate placement under the lock)
      BB_LOG(ERROR) << "scrub: digest mismatch on " << cd.key << " @ "
                    << bad_sh->pool_id << "+" << bad_sh->offset << " (want "
                    << bad_want << " got " << bad_got << ")";
      bool dropped = false;
      bool object_gone = false;
      {
        std::unique_lock lk(objects_mu_);
        auto it = objects_.find(cd.key);
        if (it == objects_.end()) continue;
        auto& copies = it->second.copies;
        for (auto cit = copies.begin(); cit != copies.end(); ++cit) {
          if (cit->shards == shards) {
            copies.erase(cit);
            bump_placement_epoch_locked();
            dropped = true;
            break;
          }
        }
        if (!dropped) continue;  // placement changed under us
        if (copies.empty()) {
          // last copy was corrupt: the object is lost — drop it rather
          // than serve bytes that cannot verify
          BB_LOG(ERROR) << "scrub: all copies of " << cd.key
                        << " corrupt — removing object";
          remove_object_locked(cd.key);
          object_gone = true;
        } else {
          mark_dirty_locked(cd.key, false);
          bump_view();
        }
      }
      if (dropped && !object_gone) allocator_.free_ranges(cd.key, shards);
      ctr_scrubbed_.fetch_add(1);
      ++quarantined;
    }
  }
  return quarantined;
}

void KeystoneService::run_compaction_once() {
  const double thr = config_.compact_fragmentation_threshold;
  if (thr <= 0) return;
  for (const auto& p : allocator_.pools()) {
    auto st = allocator_.pool_stats(p.pool_id);
    if (!st.ok() || st.value().used == 0) continue;
    if (st.value().fragmentation <= thr) continue;
    auto moved = compact_pool(p.pool_id, config_.tier_max_moves_per_cycle);
    if (moved.ok() && moved.value() > 0)
      BB_LOG(INFO) << "auto-compacted " << p.pool_id << ": moved "
                   << moved.value() << " objects (frag "
                   << st.value().fragmentation << ")";
  }
}

// ------------------------------------------------------------- watchers

void KeystoneService::load_existing_state() {
  auto workers = coord_->get_prefix(prefix() + "/workers/");
  if (workers.ok()) {
    for (const auto& kv : workers.value()) {
      auto w = WorkerInfo::from_json(json::parse_or_null(kv.value));
      if (!w.worker_id.empty()) {
        std::unique_lock lk(workers_mu_);
        w.last_heartbeat_ms = now_ms();
        workers_[w.worker_id] = w;
      }
    }
  }
  auto pools = coord_->get_prefix(prefix() + "/memory_pools/");
  if (pools.ok()) {
    for (const auto& kv : pools.value()) {
      auto p = MemoryPool::from_json(json::parse_or_null(kv.value));
      if (!p.pool_id.empty()) allocator_.upsert_pool(p);
    }
  }
  if (config_.persist_objects) {
    auto objs = coord_->get_prefix(prefix() + "/objects/");
    size_t restored = 0, dropped = 0;
    if (objs.ok()) {
      for (const auto& kv : objs.value()) {
        ObjectMeta meta;
        if (!serde::from_bytes(kv.value, meta) || meta.key.empty()) {
          ++dropped;
          coord_->del(kv.key);
          continue;
        }
        {
          std::shared_lock lk(objects_mu_);
          if (objects_.count(meta.key)) continue;  // already live (rescan)
        }
        auto ad = allocator_.adopt(meta.key, meta.copies);
        if (!ad.ok()) {
          // pool gone or range occupied: the bytes are unreachable
          ++dropped;
          coord_->del(kv.key);
          continue;
        }
        std::unique_lock lk(objects_mu_);
        objects_[meta.key] = std::move(meta);
        ++restored;
      }
    }
    if (restored || dropped)
      BB_LOG(INFO) << "restored " << restored << " objects from coordination ("
                   << dropped << " dropped)";
  }
  bump_view();
}

void KeystoneService::setup_watchers() {
  auto w1 = coord_->watch_prefix(prefix() + "/workers/",
                                 [this](const coord::WatchEvent& ev) {
                                   handle_worker_event(ev);
                                 });
  auto w2 = coord_->watch_prefix(prefix() + "/memory_pools/",
                                 [this](const coord::WatchEvent& ev) {
                                   handle_pool_event(ev);
                                 });
  auto w3 = coord_->watch_prefix(prefix() + "/heartbeat/",
                                 [this](const coord::WatchEvent& ev) {
                                   handle_heartbeat_event(ev);
                                 });
  for (auto& w : {w1, w2, w3})
    if (w.ok()) watch_ids_.push_back(w.value());
}

void KeystoneService::handle_worker_event(const coord::WatchEvent& ev) {
  cb_inflight_.fetch_add(1);
  struct G { std::atomic<int>& c; ~G() { c.fetch_sub(1); } } _g{cb_inflight_};
  if (!running_.load()) return;
  auto id = ev.key.substr(ev.key.rfind('/') + 1);
  if (ev.type == coord::EventType::PUT) {
    auto w = WorkerInfo::from_json(json::parse_or_null(ev.value));
    if (w.worker_id.empty()) return;
    std::unique_lock lk(workers_mu_);
    w.last_heartbeat_ms = now_ms();
    workers_[w.worker_id] = w;
    bump_view();
  } else {
    cleanup_dead_worker(id);
  }
}

void KeystoneService::handle_pool_event(const coord::WatchEvent& ev) {
  cb_inflight_.fetch_add(1);
  struct G { std::atomic<int>& c; ~G() { c.fetch_sub(1); } } _g{cb_inflight_};
  if (!running_.load()) return;
  if (ev.type == coord::EventType::PUT) {
    auto p = MemoryPool::from_json(json::parse_or_null(ev.value));
    if (p.pool_id.empty()) return;
    allocator_.upsert_pool(p);
  } else {
    auto id = ev.key.substr(ev.key.rfind('/') + 1);
    allocator_.remove_pool(id);
  }
  bump_view();
}

void KeystoneService::handle_heartbeat_event(const coord::WatchEvent& ev) {
  cb_inflight_.fetch_add(1);
  struct G { std::atomic<int>& c; ~G() { c.fetch_sub(1); } } _g{cb_inflight_};
  if (!running_.load()) return;
  auto id = ev.key.substr(ev.key.rfind('/') + 1);
  if (ev.type == coord::EventType::PUT) {
    std::unique_lock lk(workers_mu_);
    auto it = workers_.find(id);
    if (it != workers_.end()) it->second.last_heartbeat_ms = now_ms();
  } else {
    // TTL expiry or explicit delete ⇒ the worker is dead
    BB_LOG(WARN) << "worker heartbeat lost: " << id << " (event="
                 << (ev.type == coord::EventType::DELETE ? "DELETE" : "EXPIRE")
                 << " key=" << ev.key << ")";
    cleanup_dead_worker(id);
  }
}

void KeystoneService::cleanup_dead_worker(const WorkerId& id) {
  bool existed = false;
  {
    std::unique_lock lk(workers_mu_);
    existed = workers_.erase(id) > 0;
  }
  // drop the worker's pools from the placement engine
  for (const auto& p : allocator_.pools())
    if (p.worker_id == id) allocator_.remove_pool(p.pool_id);

  // Drop dead copies from object metadata so gets never return placements on
  // a dead worker (the reference served stale placements).
  {
    std::unique_lock lk(objects_mu_);
    std::vector<ObjectKey> lost;
    for (auto& [key, meta] : objects_) {
      auto& copies = meta.copies;
      auto before = copies.size();
      auto dead = std::stable_partition(copies.begin(), copies.end(),
                                        [&](const CopyPlacement& c) {
                                          for (const auto& s : c.shards)
                                            if (s.worker_id == id) return false;
                                          return true;
                                        });
      // Trim the matching leases from the allocator ledger too: if the worker
      // re-registers the same pool_id with a fresh PoolAllocator, a later
      // free/expiry must not replay stale leases into the new allocator
      // (double allocation / freeing live ranges).
      for (auto it = dead; it != copies.end(); ++it)
        allocator_.free_ranges(key, it->shards);
      if (dead != copies.end()) bump_placement_epoch_locked();
      copies.erase(dead, copies.end());
      if (copies.empty() && before > 0) lost.push_back(key);
    }
    for (const auto& k : lost) remove_object_locked(k);
    if (!lost.empty())
      BB_LOG(WARN) << "worker " << id << " death lost " << lost.size()
                   << " objects (no surviving replicas)";
  }
  // remove persistent registration keys (worker may have died without cleanup)
  coord_->del(prefix() + "/workers/" + id);
  auto pools = coord_->get_prefix(prefix() + "/memory_pools/" + id + "/");
  if (pools.ok())
    for (const auto& kv : pools.value()) coord_->del(kv.key);
  if (existed) bump_view();
}

}  // namespace blackbird
