#include "blackbird/keystone/keystone_rpc.h"

#include "blackbird/common/log.h"
#include "blackbird/rpc/methods.h"
#include "blackbird/rpc/net.h"

namespace blackbird {

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
struct StatusListMsg {
  std::vector<int32_t> statuses;
  BB_FIELDS(statuses)
};
struct ExistsListMsg {
  std::vector<uint8_t> exists;
  BB_FIELDS(exists)
};
struct WorkersInfoMsg {
  std::vector<WorkerInfo> workers;
  BB_FIELDS(workers)
};
struct PoolsMsg {
  std::vector<MemoryPool> pools;
  BB_FIELDS(pools)
};
struct PutCompleteListMsg {
  std::vector<PutCompleteRequest> reqs;
  BB_FIELDS(reqs)
};

template <typename Req>
Result<Req> decode(const std::string& body) {
  Req r{};
  if (!serde::from_bytes(body, r))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad request body"};
  return r;
}
}  // namespace

KeystoneServer::KeystoneServer(std::shared_ptr<KeystoneService> service)
    : service_(std::move(service)) {
  register_handlers();
}

KeystoneServer::~KeystoneServer() { stop(); }

void KeystoneServer::register_handlers() {
  namespace M = rpc::methods;
  using Ctx = rpc::RpcServer::ConnCtx;
  auto& ks = *service_;

  rpc_.register_handler(M::PING, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    PingResponse p;
    p.view_version = ks.get_view_version();
    p.server_time_ms = wall_ms();
    p.is_leader = ks.is_leader();
    return serde::to_bytes(p);
  });
  rpc_.register_handler(M::OBJECT_EXISTS, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeyMsg>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(BoolMsg{static_cast<uint8_t>(ks.object_exists(r->key))});
  });
  rpc_.register_handler(M::GET_WORKERS, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeyMsg>(b);
    if (!r.ok()) return r.error();
    auto resp = ks.get_workers(r->key);
    if (!resp.ok()) return resp.error();
    return serde::to_bytes(resp.value());
  });
  rpc_.register_handler(M::PUT_START, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<PutStartRequest>(b);
    if (!r.ok()) return r.error();
    auto resp = ks.put_start(r->key, r->size, r->config);
    if (!resp.ok()) return resp.error();
    return serde::to_bytes(resp.value());
  });
  rpc_.register_handler(M::PUT_COMPLETE, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<PutCompleteRequest>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(ks.put_complete(r->key, r->checksum, r->shard_digests));
    return std::string{};
  });
  rpc_.register_handler(M::PUT_CANCEL, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeyMsg>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(ks.put_cancel(r->key));
    return std::string{};
  });
  rpc_.register_handler(M::REMOVE_OBJECT, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeyMsg>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(ks.remove_object(r->key));
    return std::string{};
  });
  rpc_.register_handler(M::REMOVE_ALL_OBJECTS, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    return serde::to_bytes(U64Msg{ks.remove_all_objects()});
  });
  rpc_.register_handler(M::GET_WORKERS_INFO, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    return serde::to_bytes(WorkersInfoMsg{ks.get_workers_info()});
  });
  rpc_.register_handler(M::GET_MEMORY_POOLS, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    return serde::to_bytes(PoolsMsg{ks.get_memory_pools()});
  });
  rpc_.register_handler(M::REMOVE_WORKER, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<KeyMsg>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(ks.remove_worker(r->key));
    return std::string{};
  });
  rpc_.register_handler(M::GET_CLUSTER_STATS, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    return serde::to_bytes(ks.get_cluster_stats());
  });
  rpc_.register_handler(M::GET_VIEW_VERSION, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    return serde::to_bytes(U64Msg{ks.get_view_version()});
  });
  rpc_.register_handler(M::LIST_OBJECTS, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    serde::Dec d(b.data(), b.size());
    std::string prefix = d.str();
    uint32_t limit = d.num<uint32_t>();
    if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad list request"};
    serde::Enc e;
    serde::put(e, ks.list_objects(prefix, limit ? limit : 1000));
    return std::move(e.buf);
  });
  rpc_.register_handler(M::ADMIN_SCRUB, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    serde::Dec d(b.data(), b.size());
    uint32_t max_objects = d.num<uint32_t>();
    return serde::to_bytes(U64Msg{ks.run_scrub_once(d.ok() ? max_objects : 0)});
  });
  rpc_.register_handler(M::ADMIN_REPAIR, [&ks](const std::string&, const Ctx&) -> Result<std::string> {
    ks.run_repair_once();
    return std::string{};
  });
  rpc_.register_handler(M::ADMIN_COMPACT, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<KeyMsg>(b);  // key = pool id
    if (!r.ok()) return r.error();
    auto moved = ks.compact_pool(r->key);
    if (!moved.ok()) return moved.error();
    return serde::to_bytes(U64Msg{moved.value()});
  });
  rpc_.register_handler(M::BATCH_PUT_START, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    // standby answers with a CALL-level error (not per-item statuses) so the
    // client's failover machinery rediscovers the leader and retries
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<BatchPutStartRequest>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(ks.batch_put_start(r->requests));
  });
  rpc_.register_handler(M::BATCH_PUT_COMPLETE, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<PutCompleteListMsg>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(StatusListMsg{ks.batch_put_complete(r->reqs)});
  });
  rpc_.register_handler(M::BATCH_PUT_CANCEL, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeysMsg>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(StatusListMsg{ks.batch_put_cancel(r->keys)});
  });
  rpc_.register_handler(M::BATCH_GET_WORKERS, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeysMsg>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(ks.batch_get_workers(r->keys));
  });
  // ---- compact v2 batch protocol: pool-table + fixed-width placements ----
  rpc_.register_handler(M::BATCH_PUT_START2, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    serde::Dec d(b.data(), b.size());
    uint32_t count = d.num<uint32_t>();
    uint64_t uniform = d.num<uint64_t>();
    std::vector<PutStartRequest> reqs;
    reqs.reserve(count);
    PlacementConfig cfg{};
    std::vector<std::string> keys(count);
    std::vector<uint64_t> sizes(count, uniform);
    if (uniform == 0)
      for (uint32_t i = 0; i < count; ++i) sizes[i] = d.num<uint64_t>();
    for (uint32_t i = 0; i < count; ++i) keys[i] = d.str();
    serde::get(d, cfg);
    const uint8_t want_token = d.remaining() ? d.num<uint8_t>() : 0;
    if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad v2 request"};
    cfg.max_workers_per_copy = 1;  // v2 contract: single-shard copies
    for (uint32_t i = 0; i < count; ++i)
      reqs.push_back({std::move(keys[i]), sizes[i], cfg});
    auto resp = ks.batch_put_start(reqs);

    uint64_t token = 0;
    if (want_token) {
      bool all_ok = !resp.items.empty();
      for (auto& it : resp.items)
        if (it.status != 0) { all_ok = false; break; }
      if (all_ok) token = ks.create_put_session(reqs);
    }

    serde::Enc e;
    e.num<uint64_t>(resp.view_version);
    e.num<uint64_t>(token);  // 0 = no session granted
    // build the pool table
    std::map<std::string, uint16_t> pool_idx;
    std::vector<const std::string*> table;
    for (auto& it : resp.items)
      for (auto& c : it.copies)
        for (auto& sh : c.shards)
          if (pool_idx.emplace(sh.pool_id, static_cast<uint16_t>(table.size())).second)
            table.push_back(&sh.pool_id);
    e.num<uint16_t>(static_cast<uint16_t>(table.size()));
    for (auto* p2 : table) e.str(*p2);
    for (auto& it : resp.items) {
      e.num<uint8_t>(static_cast<uint8_t>(it.status == 0 ? 0 : 1));
      if (it.status != 0) {
        e.num<int32_t>(it.status);
        continue;
      }
      e.num<uint8_t>(static_cast<uint8_t>(it.copies.size()));
      for (auto& c : it.copies) {
        auto& sh = c.shards[0];
        e.num<uint16_t>(pool_idx[sh.pool_id]);
        e.num<uint64_t>(sh.offset);
      }
    }
    return std::move(e.buf);
  });
  rpc_.register_handler(M::BATCH_GET_WORKERS2, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    serde::Dec d(b.data(), b.size());
    uint32_t count = d.num<uint32_t>();
    std::vector<std::string> keys(count);
    for (uint32_t i = 0; i < count; ++i) keys[i] = d.str();
    if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad v2 request"};
    auto resp = ks.batch_get_workers(keys);

    serde::Enc e;
    std::map<std::string, uint16_t> pool_idx;
    std::vector<const std::string*> table;
    for (auto& it : resp.items)
      for (auto& c : it.info.copies)
        for (auto& sh : c.shards)
          if (pool_idx.emplace(sh.pool_id, static_cast<uint16_t>(table.size())).second)
            table.push_back(&sh.pool_id);
    e.num<uint16_t>(static_cast<uint16_t>(table.size()));
    for (auto* p2 : table) e.str(*p2);
    for (auto& it : resp.items) {
      // multi-shard copies cannot be encoded in v2 — signal fallback
      bool multi = false;
      for (auto& c : it.info.copies)
        if (c.shards.size() != 1) multi = true;
      if (it.status != 0 || multi) {
        e.num<uint8_t>(1);
        e.num<int32_t>(it.status != 0
                           ? it.status
                           : static_cast<int32_t>(ErrorCode::NOT_IMPLEMENTED));
        continue;
      }
      e.num<uint8_t>(0);
      e.num<uint64_t>(it.info.size);
      e.num<uint64_t>(it.info.checksum);
      e.num<uint8_t>(static_cast<uint8_t>(it.info.copies.size()));
      for (auto& c : it.info.copies) {
        e.num<uint16_t>(pool_idx[c.shards[0].pool_id]);
        e.num<uint64_t>(c.shards[0].offset);
      }
    }
    return std::move(e.buf);
  });
  rpc_.register_handler(M::BATCH_UPSERT_START, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    serde::Dec d(b.data(), b.size());
    uint64_t token = d.num<uint64_t>();
    if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad upsert request"};
    BB_RETURN_IF_ERROR(ks.upsert_start_token(token));
    return std::string{};
  });
  rpc_.register_handler(M::BATCH_COMMIT_TOKEN, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    serde::Dec d(b.data(), b.size());
    uint64_t token = d.num<uint64_t>();
    uint8_t flags = d.num<uint8_t>();  // bit 0: release session after commit
    uint32_t n = d.num<uint32_t>();
    if (!d.ok() || d.remaining() != n * sizeof(uint64_t) || n > (1u << 24))
      return Error{ErrorCode::PROTOCOL_ERROR, "bad commit request"};
    std::vector<uint64_t> digests(n);
    for (uint32_t i = 0; i < n; ++i) digests[i] = d.num<uint64_t>();
    BB_RETURN_IF_ERROR(ks.commit_token(token, digests, flags & 1));
    return std::string{};
  });
  rpc_.register_handler(M::BATCH_REMOVE, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeysMsg>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(StatusListMsg{ks.batch_remove(r->keys)});
  });
  rpc_.register_handler(M::BATCH_OBJECT_EXISTS, [&ks](const std::string& b, const Ctx&) -> Result<std::string> {
    if (!ks.is_leader())
      return Error{ErrorCode::NOT_LEADER, "standby keystone; retry on leader"};
    auto r = decode<KeysMsg>(b);
    if (!r.ok()) return r.error();
    return serde::to_bytes(ExistsListMsg{ks.batch_object_exists(r->keys)});
  });
}

Result<void> KeystoneServer::start() {
  const auto& addr = service_->config().listen_address;
  if (net::is_unix_endpoint(addr)) {
    BB_RETURN_IF_ERROR(rpc_.start(addr, 0));
    service_->set_advertised_endpoint(addr);
    BB_LOG(INFO) << "keystone RPC listening on " << addr;
    if (!service_->config().metrics_address.empty()) {
      metrics_ = std::make_unique<MetricsHttpServer>(*service_);
      auto mr = metrics_->start(service_->config().metrics_address);
      if (!mr.ok())
        BB_LOG(WARN) << "metrics server failed to start: " << mr.message();
    }
    return {};
  }
  auto hp = net::split_endpoint(addr);
  if (!hp.ok()) return hp.error();
  BB_RETURN_IF_ERROR(rpc_.start(hp.value().first, hp.value().second));
  service_->set_advertised_endpoint(rpc_.endpoint());
  BB_LOG(INFO) << "keystone RPC listening on " << rpc_.endpoint();
  if (!service_->config().metrics_address.empty()) {
    metrics_ = std::make_unique<MetricsHttpServer>(*service_);
    auto mr = metrics_->start(service_->config().metrics_address);
    if (!mr.ok())
      BB_LOG(WARN) << "metrics server failed to start: " << mr.message();
  }
  return {};
}

void KeystoneServer::stop() {
  if (metrics_) metrics_->stop();
  rpc_.stop();
}

Result<std::shared_ptr<KeystoneServer>> create_and_start_keystone(
    const KeystoneConfig& config, std::shared_ptr<coord::CoordService> coord) {
  auto svc = std::make_shared<KeystoneService>(config, std::move(coord));
  BB_RETURN_IF_ERROR(svc->initialize());
  BB_RETURN_IF_ERROR(svc->start());
  auto server = std::make_shared<KeystoneServer>(svc);
  auto r = server->start();
  if (!r.ok()) {
    svc->stop();
    return r.error();
  }
  return server;
}

}  // namespace blackbird
