#include "blackbird/worker/worker_service.h"

#include "blackbird/common/log.h"
#include "blackbird/rpc/methods.h"
#include "blackbird/rpc/net.h"

namespace blackbird {

namespace {
struct WriteReq {
  std::string pool_id;
  uint64_t offset = 0;
  std::string data;  // serde bytes
  void enc(serde::Enc& e) const {
    e.str(pool_id);
    e.num(offset);
    e.bytes(data.data(), data.size());
  }
  void dec(serde::Dec& d) {
    pool_id = d.str();
    offset = d.num<uint64_t>();
    data = d.bytes();
  }
};
struct ReadReq {
  std::string pool_id;
  uint64_t offset = 0;
  uint64_t length = 0;
  BB_FIELDS(pool_id, offset, length)
};
struct ChecksumReq {
  std::string pool_id;
  uint64_t offset = 0;
  uint64_t length = 0;
  BB_FIELDS(pool_id, offset, length)
};
struct U64Msg {
  uint64_t v = 0;
  BB_FIELDS(v)
};
struct BatchWriteReq {
  std::vector<WriteReq> writes;
  BB_FIELDS(writes)
};
struct BatchReadReq {
  std::vector<ReadReq> reads;
  BB_FIELDS(reads)
};
struct PullReq {
  std::string dst_pool;
  uint64_t dst_offset = 0;
  uint64_t total_len = 0;
  std::vector<ShardPlacement> srcs;
  BB_FIELDS(dst_pool, dst_offset, total_len, srcs)
};

template <typename Req>
Result<Req> decode(const std::string& body) {
  Req r{};
  if (!serde::from_bytes(body, r))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad request body"};
  return r;
}
}  // namespace

WorkerService::WorkerService(WorkerConfig config,
                             std::shared_ptr<coord::CoordService> coord)
    : config_(std::move(config)), coord_(std::move(coord)) {
  if (!coord_) coord_ = coord::make_coord(config_.coord_endpoint);
  if (config_.worker_id.empty()) config_.worker_id = "worker-unnamed";
  if (config_.node_id.empty()) config_.node_id = net::advertise_host();
  register_handlers();
}

WorkerService::~WorkerService() { stop(); }

StorageBackend* WorkerService::backend(const PoolId& id) {
  auto it = backends_.find(id);
  return it == backends_.end() ? nullptr : it->second.get();
}

Result<void> WorkerService::initialize() {
  for (const auto& pc : config_.pools) {
    auto b = create_storage_backend(pc, config_.worker_id);
    if (!b.ok()) return b.error();
    auto init = b.value()->initialize();
    if (!init.ok())
      return Error{init.code(),
                   "pool " + pc.pool_id + ": " + init.message()};
    LocalPools::inst().add(pc.pool_id, b.value()->base_ptr(),
                           b.value()->capacity(),
                           pc.storage_class == StorageClass::RAM_GPU,
                           pc.gpu_device_id, b.value().get());
    backends_[pc.pool_id] = std::move(b.value());
  }
  auto hp = net::split_endpoint(config_.data_listen_address);
  if (!hp.ok()) return hp.error();
  BB_RETURN_IF_ERROR(data_rpc_.start(hp.value().first, hp.value().second));
  BB_LOG(INFO) << "worker " << config_.worker_id << " data plane on "
               << data_rpc_.endpoint() << " (" << backends_.size() << " pools)";
  return {};
}

std::vector<MemoryPool> WorkerService::pool_descriptors() {
  std::vector<MemoryPool> out;
  for (const auto& [id, b] : backends_) {
    MemoryPool p;
    p.pool_id = id;
    p.worker_id = config_.worker_id;
    p.node_id = config_.node_id;
    p.storage_class = b->storage_class();
    p.size = b->capacity();
    p.used = b->stats().used;
    p.access = b->access_info();
    p.access.endpoint = data_rpc_.endpoint();
    out.push_back(std::move(p));
  }
  return out;
}

void WorkerService::register_with_coord() {
  WorkerInfo info;
  info.worker_id = config_.worker_id;
  info.node_id = config_.node_id;
  info.data_endpoint = data_rpc_.endpoint();
  info.registered_ms = wall_ms();
  coord_->put(prefix() + "/workers/" + config_.worker_id,
              info.to_json().dump(), 0);
  for (const auto& p : pool_descriptors())
    coord_->put(prefix() + "/memory_pools/" + config_.worker_id + "/" + p.pool_id,
                p.to_json().dump(), 0);
}

Result<void> WorkerService::start() {
  if (running_.exchange(true)) return {};
  if (coord_) {
    if (auto* cc = dynamic_cast<coord::CoordClient*>(coord_.get())) {
      cc->set_on_reconnect([this] {
        if (!running_.load()) return;  // teardown already under way
        BB_LOG(WARN) << "coordination restarted — re-registering worker "
                     << config_.worker_id;
        register_with_coord();
      });
    }
    register_with_coord();
    heartbeat_thread_ = std::thread([this] { heartbeat_loop(); });
  }
  return {};
}

void WorkerService::stop() {
  if (!running_.exchange(false)) return;
  {
    std::lock_guard<std::mutex> g(hb_mu_);  // no lost wakeup on stop
  }
  hb_cv_.notify_all();
  if (heartbeat_thread_.joinable()) heartbeat_thread_.join();
  if (coord_) {
    coord_->del(prefix() + "/heartbeat/" + config_.worker_id);
    coord_->del(prefix() + "/workers/" + config_.worker_id);
    for (const auto& [id, b] : backends_)
      coord_->del(prefix() + "/memory_pools/" + config_.worker_id + "/" + id);
  }
  data_rpc_.stop();
  for (auto& [id, b] : backends_) {
    LocalPools::inst().remove(id);
    b->shutdown();
  }
}

void WorkerService::heartbeat_loop() {
  const std::string key = prefix() + "/heartbeat/" + config_.worker_id;
  while (running_) {
    auto r = coord_->put(key, std::to_string(wall_ms()), config_.heartbeat_ttl_ms);
    if (!r.ok()) BB_LOG(WARN) << "heartbeat failed: " << r.message();
    std::unique_lock<std::mutex> lk(hb_mu_);
    hb_cv_.wait_for(lk, std::chrono::milliseconds(config_.heartbeat_interval_ms),
                    [this] { return !running_.load(); });
  }
}

void WorkerService::register_handlers() {
  namespace M = rpc::methods;
  using Ctx = rpc::RpcServer::ConnCtx;

  data_rpc_.register_handler(M::DATA_WRITE, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<WriteReq>(b);
    if (!r.ok()) return r.error();
    auto* be = backend(r->pool_id);
    if (!be) return Error{ErrorCode::POOL_NOT_FOUND, r->pool_id};
    BB_RETURN_IF_ERROR(be->write(r->offset, r->data.data(), r->data.size()));
    return std::string{};
  });
  data_rpc_.register_handler(M::DATA_READ, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<ReadReq>(b);
    if (!r.ok()) return r.error();
    auto* be = backend(r->pool_id);
    if (!be) return Error{ErrorCode::POOL_NOT_FOUND, r->pool_id};
    // Bound the attacker-controlled length BEFORE allocating the response:
    // oversized requests must not zero-fill gigabytes (or bad_alloc) only to
    // be rejected by the backend, nor exceed the frame limit at send time.
    if (r->length > rpc::kMaxFrame || r->length > be->capacity())
      return Error{ErrorCode::INVALID_OFFSET, "read length out of bounds"};
    std::string out;
    out.resize(r->length);
    BB_RETURN_IF_ERROR(be->read(r->offset, out.data(), r->length));
    return out;  // raw payload (no serde wrapper) — caller knows the length
  });
  data_rpc_.register_handler(M::DATA_BATCH_WRITE, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<BatchWriteReq>(b);
    if (!r.ok()) return r.error();
    for (const auto& w : r->writes) {
      auto* be = backend(w.pool_id);
      if (!be) return Error{ErrorCode::POOL_NOT_FOUND, w.pool_id};
      BB_RETURN_IF_ERROR(be->write(w.offset, w.data.data(), w.data.size()));
    }
    return std::string{};
  });
  data_rpc_.register_handler(M::DATA_BATCH_READ, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<BatchReadReq>(b);
    if (!r.ok()) return r.error();
    serde::Enc e;
    e.num<uint32_t>(static_cast<uint32_t>(r->reads.size()));
    uint64_t total = 0;
    std::string tmp;
    for (const auto& rd : r->reads) {
      auto* be = backend(rd.pool_id);
      if (!be) return Error{ErrorCode::POOL_NOT_FOUND, rd.pool_id};
      total += rd.length;
      if (rd.length > be->capacity() || total > rpc::kMaxFrame)
        return Error{ErrorCode::INVALID_OFFSET, "batch read length out of bounds"};
      tmp.resize(rd.length);
      BB_RETURN_IF_ERROR(be->read(rd.offset, tmp.data(), rd.length));
      e.bytes(tmp.data(), tmp.size());
    }
    return std::move(e.buf);
  });
  data_rpc_.register_handler(M::DATA_CHECKSUM, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<ChecksumReq>(b);
    if (!r.ok()) return r.error();
    auto* be = backend(r->pool_id);
    if (!be) return Error{ErrorCode::POOL_NOT_FOUND, r->pool_id};
    auto cs = be->checksum(r->offset, r->length);
    if (!cs.ok()) return cs.error();
    return serde::to_bytes(U64Msg{cs.value()});
  });
  data_rpc_.register_handler(M::DATA_PULL, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<PullReq>(b);
    if (!r.ok()) return r.error();
    auto* be = backend(r->dst_pool);
    if (!be) return Error{ErrorCode::POOL_NOT_FOUND, r->dst_pool};
    uint64_t sum = 0;
    for (const auto& s2 : r->srcs) sum += s2.length;
    if (sum != r->total_len)
      return Error{ErrorCode::SIZE_MISMATCH, "pull ranges do not cover object"};
    BB_RETURN_IF_ERROR(transfer_.pull(*be, r->dst_offset, r->srcs));
    return std::string{};
  });
  data_rpc_.register_handler(M::DATA_STATS, [this](const std::string&, const Ctx&) -> Result<std::string> {
    return stats_json();
  });
}

std::string WorkerService::stats_json() {
  json::Value v;
  v["worker_id"] = config_.worker_id;
  v["node_id"] = config_.node_id;
  v["data_endpoint"] = data_rpc_.endpoint();
  json::Array pools;
  for (const auto& [id, b] : backends_) {
    auto s = b->stats();
    json::Value pv;
    pv["pool_id"] = id;
    pv["storage_class"] = std::string(to_string(b->storage_class()));
    pv["capacity"] = s.capacity;
    pv["used"] = s.used;
    pv["reserved"] = s.reserved;
    pv["num_shards"] = s.num_shards;
    pools.push_back(std::move(pv));
  }
  v["pools"] = std::move(pools);
  return v.dump();
}

}  // namespace blackbird
