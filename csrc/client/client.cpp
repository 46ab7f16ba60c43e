#include "blackbird/client/client.h"

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <atomic>
#include <cstring>
#include <future>
#include <thread>

#include <hip/hip_runtime_api.h>

#include "blackbird/client/pool_mapper.h"
#include "blackbird/common/hex.h"
#include "blackbird/coord/coord.h"
#include "blackbird/common/log.h"
#include "blackbird/gpu/gpu_kernels.h"
#include "blackbird/rpc/methods.h"
#include "blackbird/worker/storage_backend.h"

namespace blackbird {

namespace M = rpc::methods;

namespace {
struct KeyMsg {
  std::string key;
  BB_FIELDS(key)
};
struct KeysMsg {
  std::vector<std::string> keys;
  BB_FIELDS(keys)
};
struct BoolMsg {
  uint8_t v = 0;
  BB_FIELDS(v)
};
struct U64Msg {
  uint64_t v = 0;
  BB_FIELDS(v)
};
struct PoolsMsg {
  std::vector<MemoryPool> pools;
  BB_FIELDS(pools)
};
struct WorkersInfoMsg {
  std::vector<WorkerInfo> workers;
  BB_FIELDS(workers)
};
struct PutCompleteListMsg {
  std::vector<PutCompleteRequest> reqs;
  BB_FIELDS(reqs)
};
struct StatusListMsg {
  std::vector<int32_t> statuses;
  BB_FIELDS(statuses)
};
struct WriteReq {
  std::string pool_id;
  uint64_t offset = 0;
  const void* src = nullptr;  // encode-only
  uint64_t len = 0;
  void enc(serde::Enc& e) const {
    e.str(pool_id);
    e.num(offset);
    e.bytes(src, len);
  }
  void dec(serde::Dec&) {}
};
struct ReadReq {
  std::string pool_id;
  uint64_t offset = 0;
  uint64_t length = 0;
  BB_FIELDS(pool_id, offset, length)
};
struct BatchWriteEnc {  // encodes to the worker's BatchWriteReq wire format
  std::vector<WriteReq> writes;
  void enc(serde::Enc& e) const {
    e.num<uint32_t>(static_cast<uint32_t>(writes.size()));
    for (const auto& w : writes) w.enc(e);
  }
  void dec(serde::Dec&) {}
};
struct BatchReadEnc {
  std::vector<ReadReq> reads;
  BB_FIELDS(reads)
};
}  // namespace

// -------------------------------------------------------------- Client

Client::Client(ClientOptions opts)
    : opts_(std::move(opts)), mapper_(std::make_shared<PoolMapper>()) {}

Client::~Client() { close(); }

Result<void> Client::connect() {
  if (opts_.keystone_endpoint.empty()) {
    if (opts_.coord_endpoint.empty())
      return Error{ErrorCode::ENDPOINT_INVALID,
                   "need keystone_endpoint or coord_endpoint"};
    // bootstrap from the coordination service registry (HA: whichever
    // keystone holds the lease re-registers under this prefix, so a fresh
    // connect always finds the current leader)
    coord::CoordClient cc;
    BB_RETURN_IF_ERROR(cc.connect(opts_.coord_endpoint));
    auto reg = cc.get_prefix("/blackbird/services/blackbird-keystone/");
    // with HA pairs, prefer the instance holding the election lease — the
    // registry lists standbys too
    auto leader = cc.get("/blackbird/clusters/" + opts_.cluster_id + "/leader");
    cc.close();
    if (!reg.ok()) return reg.error();
    if (reg.value().empty())
      return Error{ErrorCode::COORD_UNAVAILABLE,
                   "no keystone registered in coordination"};
    opts_.keystone_endpoint = reg.value()[0].value;
    if (leader.ok()) {
      for (const auto& kv : reg.value()) {
        if (kv.key.size() >= leader.value().size() &&
            kv.key.compare(kv.key.size() - leader.value().size(),
                           leader.value().size(), leader.value()) == 0) {
          opts_.keystone_endpoint = kv.value;
          break;
        }
      }
    }
  }
  return meta_.connect(opts_.keystone_endpoint);
}

Result<std::string> Client::meta_call_raw(uint16_t m, const std::string& body,
                                          int timeout_ms) {
  if (timeout_ms <= 0) timeout_ms = opts_.rpc_timeout_ms;
  auto r = meta_.call_raw(m, body, timeout_ms);
  // ~6 s total budget: leader election + registry TTL convergence after an
  // unclean leader death take a few seconds
  for (int attempt = 0; attempt < 20 && !r.ok(); ++attempt) {
    const bool standby = r.code() == ErrorCode::NOT_LEADER;
    switch (r.code()) {
      case ErrorCode::NOT_LEADER:
      case ErrorCode::NOT_CONNECTED:
      case ErrorCode::CONNECT_FAILED:
      case ErrorCode::CONNECTION_CLOSED:
      case ErrorCode::SEND_FAILED:
      case ErrorCode::RECV_FAILED:
        break;
      default:
        return r;  // not a failover condition
    }
    // a standby answer with no discovery path configured cannot improve
    if (standby && opts_.coord_endpoint.empty()) return r;
    std::this_thread::sleep_for(std::chrono::milliseconds(
        std::min(50 * (attempt + 1), standby ? 500 : 400)));
    {
      std::lock_guard<std::mutex> g(reconnect_mu_);
      if (standby || !meta_.connected()) {
        if (!opts_.coord_endpoint.empty())
          opts_.keystone_endpoint.clear();  // force re-discovery
        auto rc = connect();
        if (!rc.ok()) {
          // a dead (or not-yet-elected) leader: keep re-discovering — the
          // registry TTL drops the dead instance and the standby registers
          if (!opts_.coord_endpoint.empty()) opts_.keystone_endpoint.clear();
          continue;
        }
        reconnect_gen_.fetch_add(1);
      }
    }
    r = meta_.call_raw(m, body, timeout_ms);
  }
  return r;
}

void Client::close() {
  meta_.close();
  std::lock_guard<std::mutex> g(data_mu_);
  data_clients_.clear();
}

rpc::RpcClient* Client::data_client(const std::string& endpoint) {
  std::lock_guard<std::mutex> g(data_mu_);
  auto it = data_clients_.find(endpoint);
  if (it != data_clients_.end()) return it->second.get();
  auto c = std::make_unique<rpc::RpcClient>();
  if (!c->connect(endpoint).ok()) return nullptr;
  return data_clients_.emplace(endpoint, std::move(c)).first->second.get();
}

Result<AccessInfo> Client::pool_access(const PoolId& id) {
  {
    std::lock_guard<std::mutex> g(pool_cache_mu_);
    auto it = pool_cache_.find(id);
    if (it != pool_cache_.end()) return it->second;
  }
  auto pools = memory_pools();
  if (!pools.ok()) return pools.error();
  std::lock_guard<std::mutex> g(pool_cache_mu_);
  pool_cache_.clear();
  for (auto& p : pools.value()) pool_cache_[p.pool_id] = p.access;
  auto it = pool_cache_.find(id);
  if (it == pool_cache_.end())
    return Error{ErrorCode::POOL_NOT_FOUND, id};
  return it->second;
}

// ------------------------------------------------------ shard transfer

Result<void> Client::write_shard(const ShardPlacement& s, const void* src) {
  if (opts_.force_tcp) {
    AccessInfo a2;
    if (s.access.endpoint.empty()) {
      auto r = pool_access(s.pool_id);
      if (!r.ok()) return r.error();
      a2 = std::move(r.value());
    } else a2 = s.access;
    auto* dc = data_client(a2.endpoint);
    if (!dc) return Error{ErrorCode::CONNECT_FAILED, "data plane " + a2.endpoint};
    WriteReq req;
    req.pool_id = s.pool_id;
    req.offset = s.offset;
    req.src = src;
    req.len = s.length;
    auto r = dc->call_raw(M::DATA_WRITE, serde::to_bytes(req), opts_.rpc_timeout_ms);
    if (!r.ok()) return r.error();
    return {};
  }
  AccessInfo resolved;
  if (s.access.endpoint.empty()) {
    auto r = pool_access(s.pool_id);
    if (!r.ok()) return r.error();
    resolved = std::move(r.value());
  }
  const AccessInfo& a = s.access.endpoint.empty() ? resolved : s.access;
  {
    bool is_dev = false;
    if (void* base = LocalPools::inst().lookup(s.pool_id, &is_dev)) {
      uint8_t* dst = static_cast<uint8_t*>(base) + s.offset;
      if (!is_dev) {
        std::memcpy(dst, src, s.length);
        return {};
      }
      if (gpu::copy_sync(dst, src, s.length, hipMemcpyHostToDevice).ok())
        return {};
    } else if (auto* be = LocalPools::inst().backend(s.pool_id)) {
      // unmapped same-process pool (direct-IO NVMe tier): call the backend
      // directly instead of framing the bytes through TCP loopback
      return be->write(s.offset, src, s.length);
    }
  }
  if (a.kind == AccessKind::SHM && !a.shm_name.empty()) {
    // one-sided host fast path: the pool end address is not known here; map
    // generously (offset+length) — the segment is fixed-size, mapping is
    // cached by name with the first size seen, so map the whole pool lazily
    // via /dev/shm stat.
    if (void* base = mapper_->map_shm(a.shm_name, 0)) {
      std::memcpy(static_cast<uint8_t*>(base) + s.offset, src, s.length);
      return {};
    }
  }
  if (a.kind == AccessKind::HIP_IPC && !a.ipc_handle_hex.empty() &&
      gpu::available()) {
    if (void* base = mapper_->open_ipc(a.ipc_handle_hex, a.device_id)) {
      auto e = gpu::copy_sync(static_cast<uint8_t*>(base) + s.offset, src,
                              s.length, hipMemcpyHostToDevice);
      if (e.ok()) return {};
      BB_LOG(WARN) << "IPC write failed: " << e.error().message
                   << " — falling back to TCP";
    }
  }
  // TCP fallback
  auto* dc = data_client(a.endpoint);
  if (!dc) return Error{ErrorCode::CONNECT_FAILED, "data plane " + a.endpoint};
  WriteReq req;
  req.pool_id = s.pool_id;
  req.offset = s.offset;
  req.src = src;
  req.len = s.length;
  auto r = dc->call_raw(M::DATA_WRITE, serde::to_bytes(req), opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  return {};
}

Result<void> Client::read_shard(const ShardPlacement& s, void* dst) {
  if (opts_.force_tcp) {
    AccessInfo a2;
    if (s.access.endpoint.empty()) {
      auto r = pool_access(s.pool_id);
      if (!r.ok()) return r.error();
      a2 = std::move(r.value());
    } else a2 = s.access;
    auto* dc = data_client(a2.endpoint);
    if (!dc) return Error{ErrorCode::CONNECT_FAILED, "data plane " + a2.endpoint};
    ReadReq req{s.pool_id, s.offset, s.length};
    auto r = dc->call_raw(M::DATA_READ, serde::to_bytes(req), opts_.rpc_timeout_ms);
    if (!r.ok()) return r.error();
    if (r.value().size() != s.length)
      return Error{ErrorCode::SIZE_MISMATCH, "short read"};
    std::memcpy(dst, r.value().data(), s.length);
    return {};
  }
  AccessInfo resolved;
  if (s.access.endpoint.empty()) {
    auto r = pool_access(s.pool_id);
    if (!r.ok()) return r.error();
    resolved = std::move(r.value());
  }
  const AccessInfo& a = s.access.endpoint.empty() ? resolved : s.access;
  {
    bool is_dev = false;
    if (void* base = LocalPools::inst().lookup(s.pool_id, &is_dev)) {
      uint8_t* src2 = static_cast<uint8_t*>(base) + s.offset;
      if (!is_dev) {
        std::memcpy(dst, src2, s.length);
        return {};
      }
      if (gpu::copy_sync(dst, src2, s.length, hipMemcpyDeviceToHost).ok())
        return {};
    } else if (auto* be = LocalPools::inst().backend(s.pool_id)) {
      // unmapped same-process pool (direct-IO NVMe tier): read through the
      // backend instead of TCP loopback
      return be->read(s.offset, dst, s.length);
    }
  }
  if (a.kind == AccessKind::SHM && !a.shm_name.empty()) {
    if (void* base = mapper_->map_shm(a.shm_name, 0)) {
      std::memcpy(dst, static_cast<uint8_t*>(base) + s.offset, s.length);
      return {};
    }
  }
  if (a.kind == AccessKind::HIP_IPC && !a.ipc_handle_hex.empty() &&
      gpu::available()) {
    if (void* base = mapper_->open_ipc(a.ipc_handle_hex, a.device_id)) {
      auto e = gpu::copy_sync(dst, static_cast<uint8_t*>(base) + s.offset,
                              s.length, hipMemcpyDeviceToHost);
      if (e.ok()) return {};
    }
  }
  auto* dc = data_client(a.endpoint);
  if (!dc) return Error{ErrorCode::CONNECT_FAILED, "data plane " + a.endpoint};
  ReadReq req{s.pool_id, s.offset, s.length};
  auto r = dc->call_raw(M::DATA_READ, serde::to_bytes(req), opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  if (r.value().size() != s.length)
    return Error{ErrorCode::SIZE_MISMATCH, "short read"};
  std::memcpy(dst, r.value().data(), s.length);
  return {};
}

Result<void> Client::write_copies(const std::vector<CopyPlacement>& copies,
                                  const void* data, uint64_t size) {
  for (const auto& copy : copies) {
    // correct running source offset per shard (reference bug fixed)
    std::vector<std::pair<const ShardPlacement*, uint64_t>> work;
    uint64_t off = 0;
    for (const auto& s : copy.shards) {
      work.emplace_back(&s, off);
      off += s.length;
    }
    if (off != size)
      return Error{ErrorCode::SIZE_MISMATCH, "placement does not cover object"};
    if (work.size() == 1) {
      BB_RETURN_IF_ERROR(write_shard(*work[0].first,
                                     static_cast<const uint8_t*>(data)));
    } else {
      std::vector<std::future<Result<void>>> futs;
      for (auto& [sp, o] : work)
        futs.push_back(std::async(std::launch::async, [this, sp, o, data] {
          return write_shard(*sp, static_cast<const uint8_t*>(data) + o);
        }));
      for (auto& f : futs) BB_RETURN_IF_ERROR(f.get());
    }
  }
  return {};
}

Result<void> Client::read_copy(const std::vector<CopyPlacement>& copies,
                               void* dst, uint64_t size) {
  Error last{ErrorCode::NO_PLACEMENT, "no copies"};
  for (const auto& copy : copies) {
    uint64_t off = 0;
    bool ok = true;
    std::vector<std::tuple<const ShardPlacement*, uint64_t>> work;
    for (const auto& s : copy.shards) {
      work.emplace_back(&s, off);
      off += s.length;
    }
    if (off != size) continue;
    if (work.size() == 1) {
      auto r = read_shard(*std::get<0>(work[0]), dst);
      if (r.ok()) return {};
      last = r.error();
      continue;
    }
    std::vector<std::future<Result<void>>> futs;
    for (auto& [sp, o] : work)
      futs.push_back(std::async(std::launch::async, [this, sp = sp, o = o, dst] {
        return read_shard(*sp, static_cast<uint8_t*>(dst) + o);
      }));
    for (auto& f : futs) {
      auto r = f.get();
      if (!r.ok()) {
        ok = false;
        last = r.error();
      }
    }
    if (ok) return {};
  }
  return last;
}

// ------------------------------------------------------------ object ops

namespace {
// Per-copy per-shard standalone digests for striped copies (empty when every
// copy is single-shard — the whole-object checksum covers those). Recorded
// at put time so the scrubber can verify each striped shard independently.
std::vector<std::vector<uint64_t>> host_shard_digests(
    const std::vector<CopyPlacement>& copies, const void* data) {
  bool any_striped = false;
  for (const auto& c : copies)
    if (c.shards.size() > 1) any_striped = true;
  if (!any_striped) return {};
  std::vector<std::vector<uint64_t>> out;
  out.reserve(copies.size());
  for (const auto& c : copies) {
    std::vector<uint64_t> ds;
    uint64_t off = 0;
    for (const auto& s : c.shards) {
      ds.push_back(gpu::checksum_cpu(
          static_cast<const uint8_t*>(data) + off, s.length));
      off += s.length;
    }
    out.push_back(std::move(ds));
  }
  return out;
}
}  // namespace

Result<void> Client::put(const ObjectKey& key, const void* data, uint64_t size,
                         const PlacementConfig& cfg) {
  const uint64_t gen = reconnect_gen_.load();
  auto r = put_once(key, data, size, cfg);
  if (r.ok() || reconnect_gen_.load() == gen ||
      !failover_retriable(static_cast<int32_t>(r.code())))
    return r;
  // leader change mid-put: the new leader never saw the PENDING object —
  // redo once from put_start
  return put_once(key, data, size, cfg);
}

Result<void> Client::put_once(const ObjectKey& key, const void* data,
                              uint64_t size, const PlacementConfig& cfg) {
  PutStartRequest req{key, size, cfg};
  auto start = meta_call<PutStartRequest, PutStartResponse>(M::PUT_START, req,
                                                             opts_.rpc_timeout_ms);
  if (!start.ok()) return start.error();

  auto xfer = write_copies(start->copies, data, size);
  if (!xfer.ok()) {
    meta_call_raw(M::PUT_CANCEL, serde::to_bytes(KeyMsg{key}), opts_.rpc_timeout_ms);
    return xfer.error();
  }
  uint64_t checksum = cfg.checksum ? gpu::checksum_cpu(data, size) : 0;
  PutCompleteRequest done_req{key, checksum, {}};
  if (cfg.checksum)
    done_req.shard_digests = host_shard_digests(start->copies, data);
  auto done = meta_call_raw(M::PUT_COMPLETE, serde::to_bytes(done_req),
                             opts_.rpc_timeout_ms);
  if (!done.ok()) return done.error();
  return {};
}

Result<std::string> Client::get(const ObjectKey& key) {
  auto meta = meta_call<KeyMsg, GetWorkersResponse>(M::GET_WORKERS, KeyMsg{key},
                                                     opts_.rpc_timeout_ms);
  if (!meta.ok()) return meta.error();
  std::string out;
  out.resize(meta->size);
  BB_RETURN_IF_ERROR(read_copy(meta->copies, out.data(), meta->size));
  if (opts_.verify_checksum_on_get && meta->checksum != 0) {
    uint64_t got = gpu::checksum_cpu(out.data(), out.size());
    if (got != meta->checksum)
      return Error{ErrorCode::CHECKSUM_MISMATCH, key};
  }
  return out;
}

Result<uint64_t> Client::get_into(const ObjectKey& key, void* dst,
                                  uint64_t capacity) {
  auto meta = meta_call<KeyMsg, GetWorkersResponse>(M::GET_WORKERS, KeyMsg{key},
                                                     opts_.rpc_timeout_ms);
  if (!meta.ok()) return meta.error();
  if (meta->size > capacity)
    return Error{ErrorCode::SIZE_MISMATCH, "buffer too small"};
  BB_RETURN_IF_ERROR(read_copy(meta->copies, dst, meta->size));
  return meta->size;
}

Result<bool> Client::exists(const ObjectKey& key) {
  auto r = meta_call<KeyMsg, BoolMsg>(M::OBJECT_EXISTS, KeyMsg{key},
                                       opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  return r->v != 0;
}

Result<void> Client::remove(const ObjectKey& key) {
  const uint64_t gen = reconnect_gen_.load();
  auto r = meta_call_raw(M::REMOVE_OBJECT, serde::to_bytes(KeyMsg{key}),
                          opts_.rpc_timeout_ms);
  if (!r.ok()) {
    // see batch_remove: NOT_FOUND after an in-call failover = already gone
    if (r.code() == ErrorCode::OBJECT_NOT_FOUND &&
        reconnect_gen_.load() != gen)
      return {};
    return r.error();
  }
  return {};
}

Result<uint64_t> Client::remove_all() {
  auto r = meta_call_raw(M::REMOVE_ALL_OBJECTS, {}, opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  U64Msg m;
  serde::from_bytes(r.value(), m);
  return m.v;
}

// ------------------------------------------------------------- batch ops

Result<std::vector<int32_t>> Client::batch_remove(
    const std::vector<ObjectKey>& keys) {
  const uint64_t gen = reconnect_gen_.load();
  auto r = meta_call<KeysMsg, StatusListMsg>(M::BATCH_REMOVE, KeysMsg{keys},
                                              opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  if (reconnect_gen_.load() != gen) {
    // a leader failover happened inside this call: the first attempt may
    // have applied + replicated before the reply was lost, so the retried
    // remove sees NOT_FOUND — the intent (key gone) is satisfied
    for (auto& st : r->statuses)
      if (st == static_cast<int32_t>(ErrorCode::OBJECT_NOT_FOUND)) st = 0;
  }
  return std::move(r->statuses);
}



bool Client::failover_retriable(int32_t st) {
  switch (static_cast<ErrorCode>(st)) {
    case ErrorCode::OBJECT_NOT_FOUND:   // new leader never saw the PENDING put
    case ErrorCode::INVALID_STATE:      // replace-put interrupted between
                                        // start and commit: the new leader
                                        // restored the PREVIOUS committed
                                        // digest, so our commit looks like a
                                        // different-content double-commit —
                                        // a full redo re-places + rewrites
    case ErrorCode::NOT_LEADER:
    case ErrorCode::NOT_CONNECTED:
    case ErrorCode::CONNECT_FAILED:
    case ErrorCode::CONNECTION_CLOSED:
    case ErrorCode::SEND_FAILED:
    case ErrorCode::RECV_FAILED:
    case ErrorCode::RPC_FAILED:
      return true;
    default:
      return false;
  }
}

Result<std::vector<int32_t>> Client::batch_put(const std::vector<PutItem>& items,
                                               const PlacementConfig& cfg,
                                               HostPutSession* sess) {
  if (auto fast = try_host_session_put(items, sess)) return std::move(*fast);
  const uint64_t gen = reconnect_gen_.load();
  auto st = batch_put_once(items, cfg, sess);
  if (!st.ok()) return st;
  if (reconnect_gen_.load() == gen) return st;  // no failover: statuses final
  // a leader change happened during the batch: redo the items whose failure
  // is failover-shaped (the new leader lost their PENDING state) exactly once
  std::vector<PutItem> redo;
  std::vector<size_t> redo_idx;
  for (size_t i = 0; i < items.size(); ++i)
    if (failover_retriable(st.value()[i])) {
      redo.push_back(items[i]);
      redo_idx.push_back(i);
    }
  if (redo.empty()) return st;
  auto st2 = batch_put_once(redo, cfg, nullptr);
  if (!st2.ok()) return st;  // keep the first answer
  for (size_t j = 0; j < redo_idx.size(); ++j)
    st.value()[redo_idx[j]] = st2.value()[j];
  return st;
}

// Token fast path for the host tier: placements unchanged since the last
// step ⇒ two tiny RPCs around direct memcpys + CPU digests. Mirrors
// GpuClient::try_session_put, including the ordering guarantee: the server
// flips the session's objects to PENDING (pinning the placements) BEFORE
// any byte is written.
std::optional<std::vector<int32_t>> Client::try_host_session_put(
    const std::vector<PutItem>& items, HostPutSession* sess) {
  if (!sess || sess->token == 0 || sess->owner != this || items.empty())
    return std::nullopt;
  const uint32_t dpi = std::max<uint32_t>(sess->descs_per_item, 1);
  if (sess->srcs.size() != items.size() ||
      sess->dsts.size() != items.size() * dpi)
    return std::nullopt;
  for (size_t i = 0; i < items.size(); ++i)
    if (sess->srcs[i] != items[i].data || sess->sizes[i] != items[i].size) {
      sess->token = 0;
      return std::nullopt;
    }
  serde::Enc e1;
  e1.num<uint64_t>(sess->token);
  auto r1 = meta_call_raw(M::BATCH_UPSERT_START, e1.buf, opts_.rpc_timeout_ms);
  if (!r1.ok()) {
    sess->token = 0;  // stale before any write: clean fallback
    return std::nullopt;
  }
  std::vector<uint64_t> digests(items.size(), 0);
  {
    std::atomic<size_t> next{0};
    const int nthreads = std::max(
        1, std::min<int>(opts_.io_threads, static_cast<int>(items.size())));
    std::vector<std::future<void>> futs;
    for (int t = 0; t < nthreads; ++t)
      futs.push_back(std::async(std::launch::async, [&] {
        for (size_t i = next.fetch_add(1); i < items.size();
             i = next.fetch_add(1)) {
          for (uint32_t k = 0; k < dpi; ++k)
            std::memcpy(sess->dsts[i * dpi + k], items[i].data,
                        items[i].size);
          digests[i] = gpu::checksum_cpu(items[i].data, items[i].size);
        }
      }));
    for (auto& f : futs) f.get();
  }
  serde::Enc e2;
  e2.num<uint64_t>(sess->token);
  e2.num<uint8_t>(0);  // keep the session
  e2.num<uint32_t>(static_cast<uint32_t>(digests.size()));
  for (uint64_t dg : digests) e2.num<uint64_t>(dg);
  auto r2 = meta_call_raw(M::BATCH_COMMIT_TOKEN, e2.buf, opts_.rpc_timeout_ms);
  if (!r2.ok()) {
    sess->token = 0;  // placements changed mid-step: full path re-places
    return std::nullopt;
  }
  host_session_steps_.fetch_add(1);
  return std::vector<int32_t>(items.size(), 0);
}

uint8_t* Client::host_pool_base(const PoolId& id, AccessInfo* access) {
  bool is_dev = false;
  if (void* b = LocalPools::inst().lookup(id, &is_dev)) {
    // device pools stay on the shard path (write_shard/read_shard hipMemcpy)
    return is_dev ? nullptr : static_cast<uint8_t*>(b);
  }
  auto a = pool_access(id);
  if (!a.ok()) return nullptr;
  uint8_t* out = nullptr;
  if (a->kind == AccessKind::SHM && !a->shm_name.empty())
    out = static_cast<uint8_t*>(mapper_->map_shm(a->shm_name, 0));
  if (access) *access = std::move(a.value());
  return out;
}

namespace {
struct HostPoolRef {
  PoolId pool_id;
  uint8_t* base = nullptr;
  AccessInfo access;
};
}  // namespace

// Host twin of GpuClient::batch_put_device_v2: compact request, pool-table
// response, memcpy into mapped pools (shard RPC fallback), digests computed
// on the fly, commit by one-shot token when the whole batch qualifies —
// BATCH_PUT_COMPLETE never re-sends keys on the common path.
Result<std::vector<int32_t>> Client::batch_put_once_v2(
    const std::vector<PutItem>& items, const PlacementConfig& cfg,
    HostPutSession* sess) {
  const bool establish =
      sess != nullptr && cfg.replace && cfg.checksum && !opts_.force_tcp;
  serde::Enc req;
  req.num<uint32_t>(static_cast<uint32_t>(items.size()));
  uint64_t uniform = items.empty() ? 1 : items[0].size;
  for (auto& it : items)
    if (it.size != uniform) { uniform = 0; break; }
  req.num<uint64_t>(uniform);
  if (uniform == 0)
    for (auto& it : items) req.num<uint64_t>(it.size);
  for (auto& it : items) req.str(it.key);
  serde::put(req, cfg);
  const bool want_token = establish || cfg.checksum;
  req.num<uint8_t>(want_token ? 1 : 0);
  auto resp = meta_call_raw(M::BATCH_PUT_START2, req.buf, opts_.rpc_timeout_ms);
  if (!resp.ok()) return resp.error();

  serde::Dec d(resp.value().data(), resp.value().size());
  d.num<uint64_t>();  // view version
  const uint64_t token = d.num<uint64_t>();
  const uint16_t npools = d.num<uint16_t>();
  std::vector<HostPoolRef> pools(npools);
  for (uint16_t i = 0; i < npools; ++i) {
    pools[i].pool_id = d.str();
    pools[i].base = host_pool_base(pools[i].pool_id, &pools[i].access);
  }

  std::vector<int32_t> statuses(items.size(), 0);
  // per-item placement list: (pool index, offset) per copy
  std::vector<std::vector<std::pair<uint16_t, uint64_t>>> placed(items.size());
  for (size_t i = 0; i < items.size() && d.ok(); ++i) {
    if (d.num<uint8_t>() != 0) {
      statuses[i] = d.num<int32_t>();
      continue;
    }
    const uint8_t ncopies = d.num<uint8_t>();
    placed[i].reserve(ncopies);
    for (uint8_t c = 0; c < ncopies; ++c) {
      const uint16_t pi = d.num<uint16_t>();
      const uint64_t off = d.num<uint64_t>();
      if (pi >= npools)
        statuses[i] = static_cast<int32_t>(ErrorCode::PROTOCOL_ERROR);
      else
        placed[i].emplace_back(pi, off);
    }
  }
  if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad v2 response"};

  // establishment bookkeeping: one dst per copy per item, all host-mapped
  const uint32_t dpi = std::max<uint32_t>(cfg.replication, 1);
  std::vector<uint8_t*> sess_dsts;
  bool all_mapped = establish;
  if (establish) {
    sess_dsts.assign(items.size() * dpi, nullptr);
    for (size_t i = 0; i < items.size() && all_mapped; ++i) {
      if (statuses[i] != 0 || placed[i].size() != dpi) {
        all_mapped = false;
        break;
      }
      for (uint32_t k = 0; k < dpi; ++k) {
        HostPoolRef& pr = pools[placed[i][k].first];
        if (!pr.base) {
          all_mapped = false;
          break;
        }
        sess_dsts[i * dpi + k] = pr.base + placed[i][k].second;
      }
    }
  }

  // transfers + digests fan out per item (digest runs while the bytes are
  // still cache-hot from the memcpy)
  std::vector<uint64_t> digests(items.size(), 0);
  {
    std::atomic<size_t> next{0};
    const int nthreads = std::max(
        1, std::min<int>(opts_.io_threads, static_cast<int>(items.size())));
    std::vector<std::future<void>> futs;
    for (int t = 0; t < nthreads; ++t)
      futs.push_back(std::async(std::launch::async, [&] {
        for (size_t i = next.fetch_add(1); i < items.size();
             i = next.fetch_add(1)) {
          if (statuses[i] != 0) continue;
          for (auto& [pi, off] : placed[i]) {
            HostPoolRef& pr = pools[pi];
            if (pr.base) {
              std::memcpy(pr.base + off, items[i].data, items[i].size);
            } else {
              ShardPlacement sp;
              sp.pool_id = pr.pool_id;
              sp.offset = off;
              sp.length = items[i].size;
              sp.access = pr.access;
              auto r = write_shard(sp, items[i].data);
              if (!r.ok()) {
                statuses[i] = static_cast<int32_t>(r.code());
                break;
              }
            }
          }
          if (cfg.checksum && statuses[i] == 0)
            digests[i] = gpu::checksum_cpu(items[i].data, items[i].size);
        }
      }));
    for (auto& f : futs) f.get();
  }

  bool all_ok = token != 0;
  for (auto st : statuses)
    if (st != 0) { all_ok = false; break; }
  const bool keep_session = all_ok && establish && all_mapped;
  bool committed = false;
  if (all_ok) {
    serde::Enc e2;
    e2.num<uint64_t>(token);
    e2.num<uint8_t>(keep_session ? 0 : 1);  // one-shot unless establishing
    e2.num<uint32_t>(static_cast<uint32_t>(digests.size()));
    for (uint64_t dg : digests) e2.num<uint64_t>(dg);
    committed =
        meta_call_raw(M::BATCH_COMMIT_TOKEN, e2.buf, opts_.rpc_timeout_ms).ok();
  }
  if (keep_session && committed) {
    sess->token = token;
    sess->descs_per_item = dpi;
    sess->dsts = std::move(sess_dsts);
    sess->srcs.clear();
    sess->sizes.clear();
    sess->srcs.reserve(items.size());
    sess->sizes.reserve(items.size());
    for (const auto& it : items) {
      sess->srcs.push_back(it.data);
      sess->sizes.push_back(it.size);
    }
    sess->owner = this;
  } else if (sess) {
    sess->token = 0;
  }
  if (!committed) {
    PutCompleteListMsg completes;
    std::vector<std::string> cancels;
    std::vector<size_t> complete_idx;
    for (size_t i = 0; i < items.size(); ++i) {
      if (statuses[i] != 0) {
        if (statuses[i] != static_cast<int32_t>(ErrorCode::OBJECT_EXISTS) &&
            statuses[i] != static_cast<int32_t>(ErrorCode::NO_SPACE))
          cancels.push_back(items[i].key);
        continue;
      }
      completes.reqs.push_back(PutCompleteRequest{items[i].key, digests[i], {}});
      complete_idx.push_back(i);
    }
    if (!completes.reqs.empty()) {
      auto r = meta_call<PutCompleteListMsg, StatusListMsg>(
          M::BATCH_PUT_COMPLETE, completes, opts_.rpc_timeout_ms);
      if (!r.ok()) return r.error();
      for (size_t j = 0; j < complete_idx.size() && j < r->statuses.size(); ++j)
        if (r->statuses[j] != 0) statuses[complete_idx[j]] = r->statuses[j];
    }
    if (!cancels.empty())
      meta_call_raw(M::BATCH_PUT_CANCEL, serde::to_bytes(KeysMsg{cancels}),
                    opts_.rpc_timeout_ms);
  }
  return statuses;
}

Result<std::vector<std::pair<int32_t, std::string>>> Client::batch_get_once_v2(
    const std::vector<ObjectKey>& keys, bool* fallback) {
  *fallback = false;
  serde::Enc req;
  req.num<uint32_t>(static_cast<uint32_t>(keys.size()));
  for (auto& k : keys) req.str(k);
  auto resp =
      meta_call_raw(M::BATCH_GET_WORKERS2, req.buf, opts_.rpc_timeout_ms);
  if (!resp.ok()) return resp.error();

  serde::Dec d(resp.value().data(), resp.value().size());
  const uint16_t npools = d.num<uint16_t>();
  std::vector<HostPoolRef> pools(npools);
  for (uint16_t i = 0; i < npools; ++i) {
    pools[i].pool_id = d.str();
    pools[i].base = host_pool_base(pools[i].pool_id, &pools[i].access);
  }

  std::vector<std::pair<int32_t, std::string>> out(keys.size());
  std::vector<uint64_t> want(keys.size(), 0);
  std::vector<std::vector<std::pair<uint16_t, uint64_t>>> copies(keys.size());
  for (size_t i = 0; i < keys.size() && d.ok(); ++i) {
    if (d.num<uint8_t>() != 0) {
      out[i].first = d.num<int32_t>();
      if (out[i].first == static_cast<int32_t>(ErrorCode::NOT_IMPLEMENTED))
        *fallback = true;  // striped object in the batch: run the v1 path
      continue;
    }
    const uint64_t size = d.num<uint64_t>();
    want[i] = d.num<uint64_t>();
    out[i].second.resize(size);
    const uint8_t ncopies = d.num<uint8_t>();
    copies[i].reserve(ncopies);
    for (uint8_t c = 0; c < ncopies; ++c) {
      const uint16_t pi = d.num<uint16_t>();
      const uint64_t off = d.num<uint64_t>();
      if (pi < npools) copies[i].emplace_back(pi, off);
    }
  }
  if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad v2 response"};
  if (*fallback) return out;

  std::atomic<size_t> next{0};
  const int nthreads = std::max(
      1, std::min<int>(opts_.io_threads, static_cast<int>(keys.size())));
  std::vector<std::future<void>> futs;
  for (int t = 0; t < nthreads; ++t)
    futs.push_back(std::async(std::launch::async, [&] {
      for (size_t i = next.fetch_add(1); i < keys.size();
           i = next.fetch_add(1)) {
        if (out[i].first != 0) continue;
        const uint64_t size = out[i].second.size();
        Error last{ErrorCode::NO_PLACEMENT, "no copies"};
        bool done = false;
        for (auto& [pi, off] : copies[i]) {
          HostPoolRef& pr = pools[pi];
          if (pr.base) {
            std::memcpy(out[i].second.data(), pr.base + off, size);
            done = true;
          } else {
            ShardPlacement sp;
            sp.pool_id = pr.pool_id;
            sp.offset = off;
            sp.length = size;
            sp.access = pr.access;
            auto r = read_shard(sp, out[i].second.data());
            if (r.ok()) done = true;
            else last = r.error();
          }
          if (done && opts_.verify_checksum_on_get && want[i] != 0) {
            if (gpu::checksum_cpu(out[i].second.data(), size) != want[i]) {
              done = false;  // corrupt copy: try the next one
              last = Error{ErrorCode::CHECKSUM_MISMATCH, keys[i]};
            }
          }
          if (done) break;
        }
        if (!done) {
          out[i].first = static_cast<int32_t>(last.code);
          out[i].second.clear();
        }
      }
    }));
  for (auto& f : futs) f.get();
  return out;
}

Result<std::vector<int32_t>> Client::batch_put_once(const std::vector<PutItem>& items,
                                               const PlacementConfig& cfg,
                                               HostPutSession* sess) {
  if (!opts_.force_tcp && cfg.max_workers_per_copy <= 1 && !items.empty())
    return batch_put_once_v2(items, cfg, sess);
  BatchPutStartRequest breq;
  breq.requests.reserve(items.size());
  for (const auto& it : items)
    breq.requests.push_back(PutStartRequest{it.key, it.size, cfg});
  auto start = meta_call<BatchPutStartRequest, BatchPutStartResponse>(
      M::BATCH_PUT_START, breq, opts_.rpc_timeout_ms);
  if (!start.ok()) return start.error();

  std::vector<int32_t> statuses(items.size(), 0);
  std::vector<uint64_t> digests(items.size(), 0);

  if (opts_.force_tcp) {
    // one DATA_BATCH_WRITE per worker endpoint instead of one RPC per shard
    std::map<std::string, BatchWriteEnc> per_ep;
    std::map<std::string, std::vector<size_t>> ep_items;
    bool grouped = true;
    for (size_t i = 0; i < items.size() && grouped; ++i) {
      auto& item = start->items[i];
      if (item.status != 0) {
        statuses[i] = item.status;
        continue;
      }
      for (const auto& copy : item.copies) {
        uint64_t off = 0;
        for (const auto& sh : copy.shards) {
          auto a = pool_access(sh.pool_id);
          if (!a.ok()) { grouped = false; break; }
          WriteReq w;
          w.pool_id = sh.pool_id;
          w.offset = sh.offset;
          w.src = static_cast<const uint8_t*>(items[i].data) + off;
          w.len = sh.length;
          per_ep[a.value().endpoint].writes.push_back(w);
          ep_items[a.value().endpoint].push_back(i);
          off += sh.length;
        }
        if (!grouped) break;
      }
    }
    if (grouped) {
      for (auto& [ep, batch] : per_ep) {
        auto* dc = data_client(ep);
        Result<std::string> r =
            dc ? dc->call_raw(M::DATA_BATCH_WRITE, serde::to_bytes(batch),
                              opts_.rpc_timeout_ms)
               : Result<std::string>(Error{ErrorCode::CONNECT_FAILED, ep});
        if (!r.ok())
          for (auto i : ep_items[ep])
            statuses[i] = static_cast<int32_t>(r.code());
      }
      if (cfg.checksum)
        for (size_t i = 0; i < items.size(); ++i)
          if (statuses[i] == 0 && start->items[i].status == 0)
            digests[i] = gpu::checksum_cpu(items[i].data, items[i].size);
      PutCompleteListMsg completes2;
      std::vector<std::string> cancels2;
      std::vector<size_t> complete_idx2;
      for (size_t i = 0; i < items.size(); ++i) {
        if (start->items[i].status != 0) continue;
        if (statuses[i] != 0) {
          cancels2.push_back(items[i].key);
          continue;
        }
        PutCompleteRequest pc{items[i].key, digests[i], {}};
        if (cfg.checksum)
          pc.shard_digests =
              host_shard_digests(start->items[i].copies, items[i].data);
        completes2.reqs.push_back(std::move(pc));
        complete_idx2.push_back(i);
      }
      if (!completes2.reqs.empty()) {
        auto r = meta_call<PutCompleteListMsg, StatusListMsg>(
            M::BATCH_PUT_COMPLETE, completes2, opts_.rpc_timeout_ms);
        if (!r.ok()) return r.error();
        // per-item commit failures (e.g. a failover lost the PENDING state)
        // must reach the caller, not vanish
        for (size_t j = 0; j < complete_idx2.size() &&
                           j < r->statuses.size(); ++j)
          if (r->statuses[j] != 0) statuses[complete_idx2[j]] = r->statuses[j];
      }
      if (!cancels2.empty())
        meta_call_raw(M::BATCH_PUT_CANCEL, serde::to_bytes(KeysMsg{cancels2}),
                       opts_.rpc_timeout_ms);
      return statuses;
    }
  }

  // transfers + digests fan out across the IO pool (each object independent)
  {
    std::atomic<size_t> next{0};
    const int nthreads =
        std::max(1, std::min<int>(opts_.io_threads, static_cast<int>(items.size())));
    std::vector<std::future<void>> futs;
    for (int t = 0; t < nthreads; ++t)
      futs.push_back(std::async(std::launch::async, [&] {
        for (size_t i = next.fetch_add(1); i < items.size();
             i = next.fetch_add(1)) {
          auto& item = start->items[i];
          if (item.status != 0) {
            statuses[i] = item.status;
            continue;
          }
          auto xfer = write_copies(item.copies, items[i].data, items[i].size);
          if (!xfer.ok()) {
            statuses[i] = static_cast<int32_t>(xfer.code());
            continue;
          }
          if (cfg.checksum)
            digests[i] = gpu::checksum_cpu(items[i].data, items[i].size);
        }
      }));
    for (auto& f : futs) f.get();
  }
  PutCompleteListMsg completes;
  std::vector<std::string> cancels;
  std::vector<size_t> complete_idx;
  for (size_t i = 0; i < items.size(); ++i) {
    if (start->items[i].status != 0) continue;  // placement failed
    if (statuses[i] != 0) {
      cancels.push_back(items[i].key);
      continue;
    }
    PutCompleteRequest pc{items[i].key, digests[i], {}};
    if (cfg.checksum)
      pc.shard_digests =
          host_shard_digests(start->items[i].copies, items[i].data);
    completes.reqs.push_back(std::move(pc));
    complete_idx.push_back(i);
  }
  if (!completes.reqs.empty()) {
    auto r = meta_call<PutCompleteListMsg, StatusListMsg>(
        M::BATCH_PUT_COMPLETE, completes, opts_.rpc_timeout_ms);
    if (!r.ok()) return r.error();
    for (size_t j = 0; j < complete_idx.size() && j < r->statuses.size(); ++j)
      if (r->statuses[j] != 0) statuses[complete_idx[j]] = r->statuses[j];
  }
  if (!cancels.empty())
    meta_call_raw(M::BATCH_PUT_CANCEL, serde::to_bytes(KeysMsg{cancels}),
                   opts_.rpc_timeout_ms);
  return statuses;
}

Result<std::vector<std::pair<int32_t, std::string>>> Client::batch_get(
    const std::vector<ObjectKey>& keys) {
  const uint64_t gen = reconnect_gen_.load();
  auto out = batch_get_once(keys);
  if (!out.ok() || reconnect_gen_.load() == gen) return out;
  // failover mid-batch: the new leader may briefly lag the persistence
  // stream — redo failover-shaped per-item failures once
  std::vector<ObjectKey> redo;
  std::vector<size_t> redo_idx;
  for (size_t i = 0; i < keys.size(); ++i)
    if (failover_retriable(out.value()[i].first)) {
      redo.push_back(keys[i]);
      redo_idx.push_back(i);
    }
  if (redo.empty()) return out;
  auto out2 = batch_get_once(redo);
  if (!out2.ok()) return out;
  for (size_t j = 0; j < redo_idx.size(); ++j)
    out.value()[redo_idx[j]] = std::move(out2.value()[j]);
  return out;
}

Result<std::vector<std::pair<int32_t, std::string>>> Client::batch_get_once(
    const std::vector<ObjectKey>& keys) {
  if (!opts_.force_tcp && !keys.empty()) {
    bool fb = false;
    auto v2 = batch_get_once_v2(keys, &fb);
    if (v2.ok() && !fb) return v2;
    if (!v2.ok()) return v2;  // whole-call errors surface (failover retries
                              // happen one level up in batch_get)
  }
  auto meta = meta_call<KeysMsg, BatchGetWorkersResponse>(
      M::BATCH_GET_WORKERS, KeysMsg{keys}, opts_.rpc_timeout_ms);
  if (!meta.ok()) return meta.error();
  std::vector<std::pair<int32_t, std::string>> out(keys.size());

  if (opts_.force_tcp) {
    // one DATA_BATCH_READ per worker endpoint
    std::map<std::string, BatchReadEnc> per_ep;
    std::map<std::string, std::vector<std::pair<size_t, uint64_t>>> slots;
    bool grouped = true;
    for (size_t i = 0; i < keys.size() && grouped; ++i) {
      auto& item = meta->items[i];
      if (item.status != 0) {
        out[i].first = item.status;
        continue;
      }
      out[i].second.resize(item.info.size);
      if (item.info.copies.empty()) {
        out[i].first = static_cast<int32_t>(ErrorCode::NO_PLACEMENT);
        continue;
      }
      uint64_t off = 0;
      for (const auto& sh : item.info.copies[0].shards) {
        auto a = pool_access(sh.pool_id);
        if (!a.ok()) { grouped = false; break; }
        per_ep[a.value().endpoint].reads.push_back(
            ReadReq{sh.pool_id, sh.offset, sh.length});
        slots[a.value().endpoint].emplace_back(i, off);
        off += sh.length;
      }
    }
    if (grouped) {
      for (auto& [ep, batch] : per_ep) {
        auto* dc = data_client(ep);
        Result<std::string> r =
            dc ? dc->call_raw(M::DATA_BATCH_READ, serde::to_bytes(batch),
                              opts_.rpc_timeout_ms)
               : Result<std::string>(Error{ErrorCode::CONNECT_FAILED, ep});
        if (!r.ok()) {
          for (auto& [i, off] : slots[ep])
            out[i].first = static_cast<int32_t>(r.code());
          continue;
        }
        serde::Dec d(r.value().data(), r.value().size());
        uint32_t n = d.num<uint32_t>();
        if (n != batch.reads.size()) {
          for (auto& [i, off] : slots[ep])
            out[i].first = static_cast<int32_t>(ErrorCode::PROTOCOL_ERROR);
          continue;
        }
        for (uint32_t j = 0; j < n; ++j) {
          auto payload = d.bytes();
          auto [i, off] = slots[ep][j];
          if (!d.ok() || payload.size() != batch.reads[j].length) {
            out[i].first = static_cast<int32_t>(ErrorCode::PROTOCOL_ERROR);
            continue;
          }
          std::memcpy(out[i].second.data() + off, payload.data(), payload.size());
        }
      }
      for (size_t i = 0; i < keys.size(); ++i)
        if (out[i].first != 0) out[i].second.clear();
      return out;
    }
  }

  std::atomic<size_t> next{0};
  const int nthreads =
      std::max(1, std::min<int>(opts_.io_threads, static_cast<int>(keys.size())));
  std::vector<std::future<void>> futs;
  for (int t = 0; t < nthreads; ++t)
    futs.push_back(std::async(std::launch::async, [&] {
      for (size_t i = next.fetch_add(1); i < keys.size();
           i = next.fetch_add(1)) {
        auto& item = meta->items[i];
        if (item.status != 0) {
          out[i].first = item.status;
          continue;
        }
        out[i].second.resize(item.info.size);
        auto r = read_copy(item.info.copies, out[i].second.data(), item.info.size);
        out[i].first = static_cast<int32_t>(r.code());
        if (!r.ok()) out[i].second.clear();
      }
    }));
  for (auto& f : futs) f.get();
  return out;
}

// ------------------------------------------------------------ cluster view

Result<ClusterStats> Client::cluster_stats() {
  auto r = meta_call_raw(M::GET_CLUSTER_STATS, {}, opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  ClusterStats s;
  if (!serde::from_bytes(r.value(), s))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad stats"};
  return s;
}

Result<std::vector<MemoryPool>> Client::memory_pools() {
  auto r = meta_call_raw(M::GET_MEMORY_POOLS, {}, opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  PoolsMsg m;
  if (!serde::from_bytes(r.value(), m))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad pools"};
  return std::move(m.pools);
}

Result<std::vector<WorkerInfo>> Client::workers_info() {
  auto r = meta_call_raw(M::GET_WORKERS_INFO, {}, opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  WorkersInfoMsg m;
  if (!serde::from_bytes(r.value(), m))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad workers"};
  return std::move(m.workers);
}

Result<std::vector<ObjectSummary>> Client::list_objects(
    const std::string& prefix, uint32_t limit) {
  serde::Enc e;
  e.str(prefix);
  e.num<uint32_t>(limit);
  auto r = meta_call_raw(M::LIST_OBJECTS, e.buf, opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  serde::Dec d(r.value().data(), r.value().size());
  std::vector<ObjectSummary> out;
  serde::get(d, out);
  if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad list"};
  return out;
}

Result<PingResponse> Client::ping() {
  auto r = meta_call_raw(M::PING, {}, opts_.rpc_timeout_ms);
  if (!r.ok()) return r.error();
  PingResponse p;
  if (!serde::from_bytes(r.value(), p))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad ping"};
  return p;
}

}  // namespace blackbird
