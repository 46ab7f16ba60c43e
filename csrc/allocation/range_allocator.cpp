#include "blackbird/allocation/range_allocator.h"

#include <algorithm>

#include "blackbird/common/log.h"

namespace blackbird {

void RangeAllocator::upsert_pool(const MemoryPool& pool) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = pools_.find(pool.pool_id);
  if (it == pools_.end()) {
    PoolState st;
    st.desc = pool;
    st.desc.used = 0;
    // disk tiers get 4 KiB granularity so shards never share an O_DIRECT
    // block (concurrent read-modify-write edges would race)
    const uint64_t align = tier_rank(pool.storage_class) >= 3 ? 4096 : 256;
    st.alloc = std::make_unique<PoolAllocator>(
        pool.size, PoolAllocator::Policy::BEST_FIT, align);
    pools_.emplace(pool.pool_id, std::move(st));
  } else {
    // keep allocator state; refresh advertised metadata (endpoint etc.)
    auto used = it->second.desc.used;
    it->second.desc = pool;
    it->second.desc.used = used;
  }
}

void RangeAllocator::remove_pool(const PoolId& id) {
  std::lock_guard<std::mutex> g(mu_);
  pools_.erase(id);
}

std::vector<MemoryPool> RangeAllocator::pools() const {
  std::lock_guard<std::mutex> g(mu_);
  std::vector<MemoryPool> out;
  out.reserve(pools_.size());
  for (const auto& [id, st] : pools_) out.push_back(st.desc);
  return out;
}

std::vector<RangeAllocator::PoolState*> RangeAllocator::candidates_locked(
    const PlacementConfig& cfg, uint64_t min_avail) {
  const auto& pref = cfg.preferred_class;
  std::vector<PoolState*> out;
  for (auto& [id, st] : pools_) {
    if (cfg.required_class && st.desc.storage_class != *cfg.required_class)
      continue;
    if (st.desc.size - st.desc.used >= min_avail) out.push_back(&st);
  }
  // Order: preferred worker first (locality hint), then preferred class,
  // then faster tier, then most-available.
  std::stable_sort(out.begin(), out.end(), [&](PoolState* a, PoolState* b) {
    if (!cfg.preferred_worker.empty()) {
      bool wa = a->desc.worker_id == cfg.preferred_worker;
      bool wb = b->desc.worker_id == cfg.preferred_worker;
      if (wa != wb) return wa;
    }
    bool pa = pref && a->desc.storage_class == *pref;
    bool pb = pref && b->desc.storage_class == *pref;
    if (pa != pb) return pa;
    int ra = tier_rank(a->desc.storage_class), rb = tier_rank(b->desc.storage_class);
    if (ra != rb) return ra < rb;
    return (a->desc.size - a->desc.used) > (b->desc.size - b->desc.used);
  });
  return out;
}

Result<CopyPlacement> RangeAllocator::allocate_one_copy_locked(
    uint64_t size, const PlacementConfig& cfg, uint32_t copy_index,
    const std::map<WorkerId, int>& worker_penalty, std::vector<Lease>& ledger) {
  const uint64_t min_shard = std::max<uint64_t>(cfg.min_shard_size, 1);
  uint32_t max_w = std::max<uint32_t>(cfg.max_workers_per_copy, 1);
  // striping only produces shards ≥ min_shard_size
  max_w = static_cast<uint32_t>(
      std::min<uint64_t>(max_w, std::max<uint64_t>(size / min_shard, 1)));

  auto cands = candidates_locked(cfg, 1);
  if (cands.empty()) return Error{ErrorCode::NO_SPACE, "no pools with capacity"};

  // Prefer workers not already used by earlier copies of this object.
  std::stable_sort(cands.begin(), cands.end(), [&](PoolState* a, PoolState* b) {
    auto ita = worker_penalty.find(a->desc.worker_id);
    auto itb = worker_penalty.find(b->desc.worker_id);
    int pa = ita == worker_penalty.end() ? 0 : ita->second;
    int pb = itb == worker_penalty.end() ? 0 : itb->second;
    return pa < pb;
  });

  for (uint32_t nw = max_w; nw >= 1; --nw) {
    // pick up to nw pools on distinct workers with enough room for one shard
    const uint64_t shard = (size + nw - 1) / nw;
    std::vector<PoolState*> picked;
    std::map<WorkerId, bool> used_worker;
    for (auto* st : cands) {
      if (picked.size() == nw) break;
      if (used_worker.count(st->desc.worker_id)) continue;
      uint64_t avail = st->desc.size - st->desc.used;
      // last shard may be smaller; require full shard for all but enough total
      if (avail < std::min<uint64_t>(shard, size)) continue;
      picked.push_back(st);
      used_worker[st->desc.worker_id] = true;
    }
    if (picked.size() < nw) continue;  // not enough distinct workers → try fewer

    // reserve round-robin: shard i ∈ pool i
    CopyPlacement copy;
    copy.copy_index = copy_index;
    std::vector<Lease> local;
    uint64_t remaining = size;
    bool failed = false;
    for (uint32_t i = 0; i < nw && remaining > 0; ++i) {
      uint64_t len = std::min<uint64_t>(shard, remaining);
      auto r = picked[i]->alloc->allocate(len);
      if (!r.ok()) { failed = true; break; }
      picked[i]->desc.used += len;
      local.push_back({picked[i]->desc.pool_id, r.value(), len});
      ShardPlacement sp;
      sp.pool_id = picked[i]->desc.pool_id;
      sp.worker_id = picked[i]->desc.worker_id;
      sp.storage_class = picked[i]->desc.storage_class;
      sp.offset = r.value();
      sp.length = len;
      // access info deliberately left empty: clients resolve pools through
      // their view-versioned pool cache; keystone enriches where it talks to
      // workers itself (migration/repair pulls)
      copy.shards.push_back(std::move(sp));
      remaining -= len;
    }
    if (failed || remaining > 0) {
      rollback_locked(local);
      continue;  // retry with fewer workers
    }
    ledger.insert(ledger.end(), local.begin(), local.end());
    return copy;
  }
  return Error{ErrorCode::NO_SPACE,
               "cannot place " + std::to_string(size) + " bytes"};
}

void RangeAllocator::rollback_locked(const std::vector<Lease>& ledger) {
  for (const auto& l : ledger) {
    auto it = pools_.find(l.pool_id);
    if (it == pools_.end()) continue;  // pool vanished (dead worker)
    it->second.alloc->free(l.offset, l.length);
    it->second.desc.used -= std::min(it->second.desc.used, l.length);
  }
}

Result<std::vector<CopyPlacement>> RangeAllocator::allocate(
    const ObjectKey& key, uint64_t size, const PlacementConfig& cfg) {
  if (size == 0) return Error{ErrorCode::INVALID_ARGUMENT, "zero-size object"};
  std::lock_guard<std::mutex> g(mu_);
  if (ledger_.count(key))
    return Error{ErrorCode::OBJECT_EXISTS, "key already allocated: " + key};

  const uint32_t replicas = std::max<uint32_t>(cfg.replication, 1);
  std::vector<CopyPlacement> copies;
  std::vector<Lease> all;
  std::map<WorkerId, int> penalty;
  for (uint32_t c = 0; c < replicas; ++c) {
    auto r = allocate_one_copy_locked(size, cfg, c, penalty, all);
    if (!r.ok()) {
      rollback_locked(all);
      return r.error();
    }
    for (const auto& sh : r.value().shards) penalty[sh.worker_id]++;
    copies.push_back(std::move(r.value()));
  }
  ledger_[key] = std::move(all);
  return copies;
}

std::vector<std::pair<int32_t, std::vector<CopyPlacement>>>
RangeAllocator::allocate_batch(const std::vector<ObjectKey>& keys,
                               const std::vector<uint64_t>& sizes,
                               const PlacementConfig& cfg) {
  std::vector<std::pair<int32_t, std::vector<CopyPlacement>>> out(keys.size());
  std::lock_guard<std::mutex> g(mu_);
  const uint32_t replicas = std::max<uint32_t>(cfg.replication, 1);
  const bool striped = cfg.max_workers_per_copy > 1;
  // pre-size the ledger so a large batch never rehashes mid-insert
  ledger_.reserve(ledger_.size() + keys.size());

  // one candidate scan+sort for the whole batch
  auto cands = candidates_locked(cfg, 1);
  // group candidates by (preferred, tier) — consecutive runs of the sort
  // key — so the batch round-robin spreads load WITHIN a tier but never
  // demotes an object to a slower tier while a faster one has room
  auto group_key = [&](PoolState* st) {
    bool pw = !cfg.preferred_worker.empty() &&
              st->desc.worker_id == cfg.preferred_worker;
    bool pref = cfg.preferred_class &&
                st->desc.storage_class == *cfg.preferred_class;
    return std::make_tuple(!pw, !pref, tier_rank(st->desc.storage_class));
  };
  std::vector<std::pair<size_t, size_t>> groups;
  for (size_t s = 0; s < cands.size();) {
    size_t e = s + 1;
    while (e < cands.size() && group_key(cands[e]) == group_key(cands[s])) ++e;
    groups.emplace_back(s, e);
    s = e;
  }
  std::vector<size_t> grr(groups.size(), 0);  // per-group round-robin cursor

  for (size_t i = 0; i < keys.size(); ++i) {
    const auto& key = keys[i];
    const uint64_t size = sizes[i];
    if (size == 0) {
      out[i].first = static_cast<int32_t>(ErrorCode::INVALID_ARGUMENT);
      continue;
    }
    if (ledger_.count(key)) {
      out[i].first = static_cast<int32_t>(ErrorCode::OBJECT_EXISTS);
      continue;
    }
    std::vector<Lease> all;
    std::vector<CopyPlacement> copies;
    bool failed = false;

    if (striped) {
      // striping path shares the per-object logic (rare in hot batches)
      std::map<WorkerId, int> penalty;
      for (uint32_t c = 0; c < replicas && !failed; ++c) {
        auto r = allocate_one_copy_locked(size, cfg, c, penalty, all);
        if (!r.ok()) { failed = true; break; }
        for (const auto& sh : r.value().shards) penalty[sh.worker_id]++;
        copies.push_back(std::move(r.value()));
      }
    } else {
      for (uint32_t c = 0; c < replicas && !failed; ++c) {
        // tier groups in order; round-robin within a group. Prefer workers
        // without a copy of this object (pass 0), fall back to any worker
        // (pass 1 — matches the soft spreading of the per-object path)
        bool placed = false;
        for (int pass = 0; pass < 2 && !placed; ++pass)
        for (size_t gi = 0; gi < groups.size() && !placed; ++gi) {
          const size_t gs = groups[gi].first;
          const size_t gn = groups[gi].second - gs;
          for (size_t t = 0; t < gn && !placed; ++t) {
            PoolState* st = cands[gs + (grr[gi] + t) % gn];
            if (st->desc.size - st->desc.used < size) continue;
            bool dup = false;
            for (const auto& cp : copies)
              for (const auto& sh : cp.shards)
                if (sh.worker_id == st->desc.worker_id) dup = true;
            if (dup && pass == 0) continue;
            auto r = st->alloc->allocate(size);
            if (!r.ok()) continue;
            st->desc.used += size;
            all.push_back({st->desc.pool_id, r.value(), size});
            CopyPlacement copy;
            copy.copy_index = c;
            ShardPlacement sp;
            sp.pool_id = st->desc.pool_id;
            sp.worker_id = st->desc.worker_id;
            sp.storage_class = st->desc.storage_class;
            sp.offset = r.value();
            sp.length = size;
            copy.shards.push_back(std::move(sp));
            copies.push_back(std::move(copy));
            grr[gi] = (grr[gi] + t + 1) % gn;
            placed = true;
          }
        }
        if (!placed) failed = true;
      }
    }

    if (failed) {
      rollback_locked(all);
      out[i].first = static_cast<int32_t>(ErrorCode::NO_SPACE);
      continue;
    }
    ledger_[key] = std::move(all);
    out[i].first = 0;
    out[i].second = std::move(copies);
  }
  return out;
}

Result<void> RangeAllocator::rename(const ObjectKey& old_key,
                                    const ObjectKey& new_key) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = ledger_.find(old_key);
  if (it == ledger_.end())
    return Error{ErrorCode::OBJECT_NOT_FOUND, old_key};
  if (ledger_.count(new_key))
    return Error{ErrorCode::OBJECT_EXISTS, new_key};
  ledger_[new_key] = std::move(it->second);
  ledger_.erase(it);
  return {};
}

Result<void> RangeAllocator::merge_into(const ObjectKey& from_key,
                                        const ObjectKey& to_key) {
  std::lock_guard<std::mutex> g(mu_);
  auto f = ledger_.find(from_key);
  if (f == ledger_.end()) return Error{ErrorCode::OBJECT_NOT_FOUND, from_key};
  auto& dst = ledger_[to_key];
  dst.insert(dst.end(), f->second.begin(), f->second.end());
  ledger_.erase(f);
  return {};
}

Result<CopyPlacement> RangeAllocator::allocate_extra_copy(
    const ObjectKey& ledger_key, uint64_t size, const PlacementConfig& cfg,
    uint32_t copy_index, const std::vector<WorkerId>& avoid_workers) {
  std::lock_guard<std::mutex> g(mu_);
  if (ledger_.count(ledger_key))
    return Error{ErrorCode::OBJECT_EXISTS, ledger_key};
  std::map<WorkerId, int> penalty;
  for (const auto& w : avoid_workers) penalty[w] += 1000;
  std::vector<Lease> leases;
  auto r = allocate_one_copy_locked(size, cfg, copy_index, penalty, leases);
  if (!r.ok()) {
    rollback_locked(leases);
    return r.error();
  }
  // hard disjointness: a repair copy on an already-holding worker adds no
  // fault tolerance — reject and wait for capacity instead
  for (const auto& sh : r.value().shards) {
    for (const auto& w : avoid_workers) {
      if (sh.worker_id == w) {
        rollback_locked(leases);
        return Error{ErrorCode::NO_SPACE,
                     "no disjoint worker for extra copy"};
      }
    }
  }
  ledger_[ledger_key] = std::move(leases);
  return r;
}

void RangeAllocator::free_batch(const std::vector<const ObjectKey*>& keys) {
  std::lock_guard<std::mutex> g(mu_);
  for (const auto* key : keys) {
    auto it = ledger_.find(*key);
    if (it == ledger_.end()) continue;
    rollback_locked(it->second);
    ledger_.erase(it);
  }
}

void RangeAllocator::free_ranges(const ObjectKey& key,
                                 const std::vector<ShardPlacement>& shards) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = ledger_.find(key);
  if (it == ledger_.end()) return;
  auto& leases = it->second;
  for (const auto& sh : shards) {
    for (auto lit = leases.begin(); lit != leases.end(); ++lit) {
      if (lit->pool_id == sh.pool_id && lit->offset == sh.offset &&
          lit->length == sh.length) {
        rollback_locked({*lit});
        leases.erase(lit);
        break;
      }
    }
  }
  if (leases.empty()) ledger_.erase(it);
}

Result<void> RangeAllocator::free(const ObjectKey& key) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = ledger_.find(key);
  if (it == ledger_.end()) return {};  // idempotent
  rollback_locked(it->second);
  ledger_.erase(it);
  return {};
}

bool RangeAllocator::can_allocate(uint64_t size, const PlacementConfig& cfg) const {
  std::lock_guard<std::mutex> g(mu_);
  const uint32_t replicas = std::max<uint32_t>(cfg.replication, 1);
  uint64_t total_avail = 0, largest = 0;
  for (const auto& [id, st] : pools_) {
    uint64_t a = st.desc.size - st.desc.used;
    total_avail += a;
    largest = std::max(largest, a);
  }
  if (total_avail < size * replicas) return false;
  uint32_t max_w = std::max<uint32_t>(cfg.max_workers_per_copy, 1);
  uint64_t min_needed = (size + max_w - 1) / max_w;
  return largest >= std::min<uint64_t>(min_needed, size);
}

Result<void> RangeAllocator::adopt(const ObjectKey& key,
                                   const std::vector<CopyPlacement>& copies) {
  std::lock_guard<std::mutex> g(mu_);
  if (ledger_.count(key)) return Error{ErrorCode::OBJECT_EXISTS, key};
  std::vector<Lease> leases;
  bool failed = false;
  for (const auto& c : copies) {
    for (const auto& sh : c.shards) {
      auto it = pools_.find(sh.pool_id);
      if (it == pools_.end()) { failed = true; break; }
      auto r = it->second.alloc->reserve_exact(sh.offset, sh.length);
      if (!r.ok()) { failed = true; break; }
      it->second.desc.used += sh.length;
      leases.push_back({sh.pool_id, sh.offset, sh.length});
    }
    if (failed) break;
  }
  if (failed) {
    rollback_locked(leases);
    return Error{ErrorCode::NO_SPACE, "adopt failed (pool gone or occupied)"};
  }
  ledger_[key] = std::move(leases);
  return {};
}

Result<AccessInfo> RangeAllocator::pool_access(const PoolId& id) const {
  std::lock_guard<std::mutex> g(mu_);
  auto it = pools_.find(id);
  if (it == pools_.end()) return Error{ErrorCode::POOL_NOT_FOUND, id};
  return it->second.desc.access;
}

Result<PoolAllocatorStats> RangeAllocator::pool_stats(const PoolId& id) const {
  std::lock_guard<std::mutex> g(mu_);
  auto it = pools_.find(id);
  if (it == pools_.end()) return Error{ErrorCode::POOL_NOT_FOUND, id};
  // merge slab freelists back into the range map first so fragmentation
  // reflects the true hole structure (maintenance path only — slabs refill
  // on demand)
  it->second.alloc->drain_slabs();
  return it->second.alloc->stats();
}

AllocatorStats RangeAllocator::stats() const {
  std::lock_guard<std::mutex> g(mu_);
  AllocatorStats s;
  s.num_pools = pools_.size();
  s.num_objects = ledger_.size();
  for (const auto& [id, st] : pools_) {
    s.total_capacity += st.desc.size;
    s.total_used += st.desc.used;
    s.fragmentation = std::max(s.fragmentation, st.alloc->stats().fragmentation);
  }
  return s;
}

}  // namespace blackbird
