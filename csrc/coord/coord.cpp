#include "blackbird/coord/coord.h"

#include <algorithm>
#include <cerrno>
#include <cstdio>
#include <cstring>
#include <unistd.h>

#include "blackbird/common/log.h"
#include "blackbird/common/serde.h"
#include "blackbird/common/types.h"

namespace blackbird::coord {

// ----------------------------------------------------------- CoordStore

CoordStore::CoordStore() {
  sweeper_ = std::thread([this] { sweeper_loop(); });
}

CoordStore::~CoordStore() {
  running_ = false;
  {
    std::lock_guard<std::mutex> g(sweep_mu_);  // no lost wakeup on stop
  }
  sweep_cv_.notify_all();
  if (sweeper_.joinable()) sweeper_.join();
}

void CoordStore::sweeper_loop() {
  while (running_) {
    {
      std::unique_lock<std::mutex> lk(sweep_mu_);
      sweep_cv_.wait_for(lk, std::chrono::milliseconds(200),
                         [this] { return !running_.load(); });
    }
    if (!running_) break;
    sweep_now();
  }
}

void CoordStore::sweep_now() {
  uint64_t now = now_ms();
  std::vector<std::string> expired;
  {
    std::lock_guard<std::mutex> g(mu_);
    for (auto it = kv_.begin(); it != kv_.end();) {
      if (it->second.deadline_ms != 0 && it->second.deadline_ms <= now) {
        expired.push_back(it->first);
        it = kv_.erase(it);
      } else {
        ++it;
      }
    }
  }
  if (!expired.empty()) dirty_ = true;
  for (const auto& k : expired) notify(EventType::EXPIRE, k, "");
}

Result<void> CoordStore::save(const std::string& path) {
  serde::Enc e;
  {
    std::lock_guard<std::mutex> g(mu_);
    e.num<uint32_t>(0xBBC05EEE);  // magic + version (v2: carries epoch)
    e.num<uint64_t>(epoch_.load());
    e.num<uint32_t>(static_cast<uint32_t>(kv_.size()));
    for (const auto& [k, entry] : kv_) {
      e.str(k);
      e.str(entry.value);
      e.num<uint64_t>(entry.deadline_ms);
    }
    dirty_ = false;
  }
  const std::string tmp = path + ".tmp";
  FILE* f = fopen(tmp.c_str(), "wb");
  if (!f) return Error{ErrorCode::INTERNAL_ERROR, "open " + tmp + ": " + strerror(errno)};
  bool ok = fwrite(e.buf.data(), 1, e.buf.size(), f) == e.buf.size();
  ok = fflush(f) == 0 && ok;
  ok = fsync(fileno(f)) == 0 && ok;
  fclose(f);
  if (!ok || rename(tmp.c_str(), path.c_str()) != 0) {
    ::unlink(tmp.c_str());
    return Error{ErrorCode::INTERNAL_ERROR, "write " + path + ": " + strerror(errno)};
  }
  return {};
}

Result<void> CoordStore::load(const std::string& path) {
  FILE* f = fopen(path.c_str(), "rb");
  if (!f) return Error{ErrorCode::KEY_NOT_FOUND, "no snapshot at " + path};
  std::string buf;
  char chunk[65536];
  size_t n;
  while ((n = fread(chunk, 1, sizeof(chunk), f)) > 0) buf.append(chunk, n);
  fclose(f);
  serde::Dec d(buf.data(), buf.size());
  const uint32_t magic = d.num<uint32_t>();
  if (magic == 0xBBC05EEE) {
    epoch_.store(d.num<uint64_t>());
  } else if (magic != 0xBBC05EED) {  // v1: no epoch field
    return Error{ErrorCode::PROTOCOL_ERROR, "bad snapshot magic in " + path};
  }
  uint32_t count = d.num<uint32_t>();
  const uint64_t now = now_ms();
  std::lock_guard<std::mutex> g(mu_);
  for (uint32_t i = 0; i < count && d.ok(); ++i) {
    std::string k = d.str();
    std::string v = d.str();
    uint64_t deadline = d.num<uint64_t>();
    if (!d.ok()) break;
    if (deadline != 0 && deadline <= now) continue;  // expired while down
    kv_[k] = Entry{std::move(v), deadline};
  }
  if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "truncated snapshot"};
  return {};
}

void CoordStore::notify(EventType t, const std::string& key, const std::string& value,
                        uint64_t ttl_ms) {
  std::vector<WatchCallback> cbs;
  {
    std::lock_guard<std::mutex> g(mu_);
    for (const auto& [id, w] : watches_) {
      if (key.rfind(w.prefix, 0) == 0) cbs.push_back(w.cb);
    }
  }
  WatchEvent ev{t, key, value, ttl_ms};
  for (auto& cb : cbs) {
    try {
      cb(ev);
    } catch (const std::exception& e) {
      BB_LOG(WARN) << "watch callback threw: " << e.what();
    }
  }
}

Result<void> CoordStore::put(const std::string& key, const std::string& value,
                             uint64_t ttl_ms) {
  {
    std::lock_guard<std::mutex> g(mu_);
    kv_[key] = Entry{value, ttl_ms ? now_ms() + ttl_ms : 0};
  }
  dirty_ = true;
  notify(EventType::PUT, key, value, ttl_ms);
  return {};
}

Result<std::string> CoordStore::get(const std::string& key) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = kv_.find(key);
  if (it == kv_.end()) return Error{ErrorCode::KEY_NOT_FOUND, key};
  if (it->second.deadline_ms != 0 && it->second.deadline_ms <= now_ms())
    return Error{ErrorCode::KEY_NOT_FOUND, key + " (expired)"};
  return it->second.value;
}

Result<void> CoordStore::del(const std::string& key) {
  bool existed = false;
  {
    std::lock_guard<std::mutex> g(mu_);
    existed = kv_.erase(key) > 0;
  }
  if (existed) {
    dirty_ = true;
    notify(EventType::DELETE, key, "");
  }
  return {};
}

Result<std::vector<KV>> CoordStore::get_prefix(const std::string& prefix) {
  std::vector<KV> out;
  uint64_t now = now_ms();
  std::lock_guard<std::mutex> g(mu_);
  for (auto it = kv_.lower_bound(prefix); it != kv_.end(); ++it) {
    if (it->first.rfind(prefix, 0) != 0) break;
    if (it->second.deadline_ms != 0 && it->second.deadline_ms <= now) continue;
    out.push_back({it->first, it->second.value});
  }
  return out;
}

Result<bool> CoordStore::cas(const std::string& key, const std::string& expected,
                             bool expect_absent, const std::string& value,
                             uint64_t ttl_ms) {
  bool won = false;
  {
    std::lock_guard<std::mutex> g(mu_);
    auto it = kv_.find(key);
    bool exists =
        it != kv_.end() && !(it->second.deadline_ms != 0 && it->second.deadline_ms <= now_ms());
    if (expect_absent) {
      won = !exists;
    } else {
      won = exists && it->second.value == expected;
    }
    if (won) kv_[key] = Entry{value, ttl_ms ? now_ms() + ttl_ms : 0};
  }
  if (won) {
    dirty_ = true;
    notify(EventType::PUT, key, value, ttl_ms);
  }
  return won;
}

Result<void> CoordStore::keep_alive(const std::string& key, uint64_t ttl_ms) {
  std::lock_guard<std::mutex> g(mu_);
  auto it = kv_.find(key);
  if (it == kv_.end()) return Error{ErrorCode::KEY_NOT_FOUND, key};
  it->second.deadline_ms = ttl_ms ? now_ms() + ttl_ms : 0;
  dirty_ = true;
  return {};
}

uint64_t CoordStore::add_watch(const std::string& prefix, WatchCallback cb) {
  uint64_t id = next_watch_++;
  std::lock_guard<std::mutex> g(mu_);
  watches_[id] = Watch{prefix, std::move(cb)};
  return id;
}

void CoordStore::remove_watch(uint64_t id) {
  std::lock_guard<std::mutex> g(mu_);
  watches_.erase(id);
}

std::vector<DumpEntry> CoordStore::dump() {
  std::vector<DumpEntry> out;
  const uint64_t now = now_ms();
  std::lock_guard<std::mutex> g(mu_);
  out.reserve(kv_.size());
  for (const auto& [k, e] : kv_) {
    if (e.deadline_ms != 0 && e.deadline_ms <= now) continue;
    out.push_back({k, e.value, e.deadline_ms ? e.deadline_ms - now : 0});
  }
  return out;
}

size_t CoordStore::size() {
  std::lock_guard<std::mutex> g(mu_);
  return kv_.size();
}

// ---------------------------------------------------------- CoordServer

namespace {
struct PutReq {
  std::string key, value;
  uint64_t ttl_ms = 0;
  BB_FIELDS(key, value, ttl_ms)
};
struct KeyReq {
  std::string key;
  BB_FIELDS(key)
};
struct PutManyReq {
  std::vector<KV> puts;
  std::vector<std::string> dels;
  BB_FIELDS(puts, dels)
};
struct ValueResp {
  std::string value;
  BB_FIELDS(value)
};
struct PrefixResp {
  std::vector<KV> kvs;
  BB_FIELDS(kvs)
};
struct CasReq {
  std::string key, expected, value;
  uint8_t expect_absent = 0;
  uint64_t ttl_ms = 0;
  BB_FIELDS(key, expected, value, expect_absent, ttl_ms)
};
struct BoolResp {
  uint8_t ok = 0;
  BB_FIELDS(ok)
};
struct KeepAliveReq {
  std::string key;
  uint64_t ttl_ms = 0;
  BB_FIELDS(key, ttl_ms)
};
struct WatchReq {
  std::string prefix;
  BB_FIELDS(prefix)
};
struct WatchResp {
  uint64_t watch_id = 0;
  BB_FIELDS(watch_id)
};
struct UnwatchReq {
  uint64_t watch_id = 0;
  BB_FIELDS(watch_id)
};

template <typename Req>
Result<Req> decode(const std::string& body) {
  Req r{};
  if (!serde::from_bytes(body, r))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad request body"};
  return r;
}

// mutation requests carry a trailing fencing epoch (u64; absent = 0,
// accepted for fencing-unaware callers — embedded/in-process paths)
template <typename Req>
Result<std::pair<Req, uint64_t>> decode_fenced(const std::string& body) {
  serde::Dec d(body.data(), body.size());
  Req r{};
  serde::get(d, r);
  if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad request body"};
  uint64_t ep = d.remaining() >= 8 ? d.num<uint64_t>() : 0;
  return std::make_pair(std::move(r), ep);
}
}  // namespace

Result<void> CoordServer::check_writable(uint64_t client_epoch) {
  const uint64_t mine = store_->epoch();
  if (client_epoch > mine) {
    // a newer leadership generation exists: this server was deposed while it
    // was down/partitioned. Fence permanently — from now on EVERY client
    // (stale ones included) gets NOT_LEADER here and rediscovers.
    if (!read_only_.exchange(true))
      BB_LOG(WARN) << "coordd fenced: local epoch " << mine
                   << " superseded by client-carried epoch " << client_epoch;
    return Error{ErrorCode::NOT_LEADER,
                 "fenced: superseded leadership (epoch=" +
                     std::to_string(client_epoch) + ")"};
  }
  if (read_only_.load())
    return Error{ErrorCode::NOT_LEADER,
                 "standby coordd (read-only) (epoch=" + std::to_string(mine) + ")"};
  // client_epoch < mine is fine: a client that has not yet observed the new
  // leadership is writing to the CURRENT leader — fencing only exists to
  // refuse the DEPOSED one (client_epoch > mine above)
  return {};
}

CoordServer::CoordServer(std::shared_ptr<CoordStore> store) : store_(std::move(store)) {
  using Ctx = rpc::RpcServer::ConnCtx;

  rpc_.register_handler(method::PUT, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode_fenced<PutReq>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(check_writable(r->second));
    BB_RETURN_IF_ERROR(store_->put(r->first.key, r->first.value, r->first.ttl_ms));
    return std::string{};
  });
  rpc_.register_handler(method::PUT_MANY, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode_fenced<PutManyReq>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(check_writable(r->second));
    for (const auto& kv : r->first.puts)
      BB_RETURN_IF_ERROR(store_->put(kv.key, kv.value, 0));
    for (const auto& k : r->first.dels)
      BB_RETURN_IF_ERROR(store_->del(k));
    return std::string{};
  });
  rpc_.register_handler(method::GET, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<KeyReq>(b);
    if (!r.ok()) return r.error();
    auto v = store_->get(r->key);
    if (!v.ok()) return v.error();
    return serde::to_bytes(ValueResp{v.value()});
  });
  rpc_.register_handler(method::DEL, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode_fenced<KeyReq>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(check_writable(r->second));
    BB_RETURN_IF_ERROR(store_->del(r->first.key));
    return std::string{};
  });
  rpc_.register_handler(method::GET_PREFIX, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<KeyReq>(b);
    if (!r.ok()) return r.error();
    auto v = store_->get_prefix(r->key);
    if (!v.ok()) return v.error();
    return serde::to_bytes(PrefixResp{std::move(v.value())});
  });
  rpc_.register_handler(method::CAS, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode_fenced<CasReq>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(check_writable(r->second));
    auto& rq = r->first;
    auto v = store_->cas(rq.key, rq.expected, rq.expect_absent != 0, rq.value, rq.ttl_ms);
    if (!v.ok()) return v.error();
    return serde::to_bytes(BoolResp{static_cast<uint8_t>(v.value() ? 1 : 0)});
  });
  rpc_.register_handler(method::KEEPALIVE, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode_fenced<KeepAliveReq>(b);
    if (!r.ok()) return r.error();
    BB_RETURN_IF_ERROR(check_writable(r->second));
    BB_RETURN_IF_ERROR(store_->keep_alive(r->first.key, r->first.ttl_ms));
    return std::string{};
  });
  rpc_.register_handler(method::WATCH, [this](const std::string& b, const Ctx& ctx) -> Result<std::string> {
    auto r = decode<WatchReq>(b);
    if (!r.ok()) return r.error();
    auto push = ctx.push;
    // The EVENT frame must carry the watch id add_watch() will hand back; a
    // shared holder lets the callback capture it before it is known.
    auto wid = std::make_shared<uint64_t>(0);
    uint64_t id = store_->add_watch(r->prefix, [push, wid](const WatchEvent& ev) {
      push(*wid, serde::to_bytes(ev));
    });
    *wid = id;
    {
      std::lock_guard<std::mutex> g(mu_);
      conn_watches_[ctx.conn_id].push_back(id);
    }
    return serde::to_bytes(WatchResp{id});
  });
  rpc_.register_handler(method::UNWATCH, [this](const std::string& b, const Ctx&) -> Result<std::string> {
    auto r = decode<UnwatchReq>(b);
    if (!r.ok()) return r.error();
    store_->remove_watch(r->watch_id);
    return std::string{};
  });

  rpc_.register_handler(method::DUMP, [this](const std::string&, const Ctx&) -> Result<std::string> {
    serde::Enc e;
    e.num<uint64_t>(store_->epoch());
    serde::put(e, store_->dump());
    return std::move(e.buf);
  });

  rpc_.register_handler(method::EPOCH, [this](const std::string&, const Ctx&) -> Result<std::string> {
    serde::Enc e;
    e.num<uint64_t>(store_->epoch());
    return std::move(e.buf);
  });

  rpc_.on_disconnect([this](uint64_t conn_id) {
    std::vector<uint64_t> ids;
    {
      std::lock_guard<std::mutex> g(mu_);
      auto it = conn_watches_.find(conn_id);
      if (it != conn_watches_.end()) {
        ids = std::move(it->second);
        conn_watches_.erase(it);
      }
    }
    for (auto id : ids) store_->remove_watch(id);
  });
}

CoordServer::~CoordServer() { stop(); }

Result<void> CoordServer::start(const std::string& host, uint16_t port) {
  return rpc_.start(host, port);
}

void CoordServer::stop() { rpc_.stop(); }

// ---------------------------------------------------------- CoordClient

CoordClient::~CoordClient() { close(); }

void CoordClient::set_on_reconnect(std::function<void()> cb) {
  std::lock_guard<std::mutex> g(mu_);
  on_reconnect_ = std::move(cb);
}

void CoordClient::install_event_callback() {
  rpc_.set_event_callback([this](uint64_t server_id, const std::string& body) {
    WatchCallback cb;
    {
      std::lock_guard<std::mutex> g(mu_);
      auto it = server_to_client_.find(server_id);
      if (it != server_to_client_.end()) {
        auto wt = watches_.find(it->second);
        if (wt != watches_.end()) cb = wt->second.cb;
      }
    }
    if (!cb) return;
    WatchEvent ev{};
    if (serde::from_bytes(body, ev)) cb(ev);
  });
}

std::string CoordClient::fenced(std::string body) {
  uint64_t e = epoch_.load();
  body.append(reinterpret_cast<const char*>(&e), sizeof(e));
  return body;
}

void CoordClient::observe_epoch(uint64_t e) {
  uint64_t cur = epoch_.load();
  while (e > cur && !epoch_.compare_exchange_weak(cur, e)) {
  }
}

// best-effort fetch of the server's fencing epoch (callers hold mu_ or are
// in connect()); never lowers the observed epoch
void CoordClient::refresh_epoch_locked() {
  auto r = rpc_.call_raw(method::EPOCH, {});
  if (!r.ok() || r.value().size() < 8) return;
  uint64_t e = 0;
  std::memcpy(&e, r.value().data(), 8);
  observe_epoch(e);
}

Result<void> CoordClient::connect(const std::string& endpoint, int timeout_ms) {
  std::vector<std::string> eps;
  size_t start = 0;
  while (start <= endpoint.size()) {
    auto comma = endpoint.find(',', start);
    if (comma == std::string::npos) comma = endpoint.size();
    auto ep = endpoint.substr(start, comma - start);
    if (!ep.empty()) eps.push_back(ep);
    start = comma + 1;
  }
  if (eps.empty()) return Error{ErrorCode::ENDPOINT_INVALID, "empty endpoint"};
  {
    std::lock_guard<std::mutex> g(mu_);
    endpoints_ = std::move(eps);
    ep_cursor_ = 0;
    timeout_ms_ = timeout_ms;
  }
  Error last{ErrorCode::CONNECT_FAILED, "unreachable"};
  for (size_t i = 0; i < endpoints_.size(); ++i) {
    auto r = rpc_.connect(endpoints_[ep_cursor_], timeout_ms);
    if (r.ok()) {
      install_event_callback();
      refresh_epoch_locked();
      return {};
    }
    last = r.error();
    ep_cursor_ = (ep_cursor_ + 1) % endpoints_.size();
  }
  return last;
}

void CoordClient::close() { rpc_.close(); }

Result<void> CoordClient::redial_locked() {
  // cycle through the endpoint list: a dead primary means the standby (if
  // configured) is next
  Error last{ErrorCode::CONNECT_FAILED, "unreachable"};
  bool ok = false;
  for (size_t i = 0; i < endpoints_.size() && !ok; ++i) {
    auto r = rpc_.connect(endpoints_[ep_cursor_], timeout_ms_);
    if (r.ok()) ok = true;
    else {
      last = r.error();
      ep_cursor_ = (ep_cursor_ + 1) % endpoints_.size();
    }
  }
  if (!ok) return last;
  install_event_callback();
  refresh_epoch_locked();
  // re-subscribe every watch on the fresh connection
  server_to_client_.clear();
  for (auto& [cid, entry] : watches_) {
    auto r = rpc_.call_raw(method::WATCH, serde::to_bytes(WatchReq{entry.prefix}));
    if (!r.ok()) return r.error();
    WatchResp resp;
    if (!serde::from_bytes(r.value(), resp))
      return Error{ErrorCode::PROTOCOL_ERROR, "bad WATCH response"};
    entry.server_id = resp.watch_id;
    server_to_client_[resp.watch_id] = cid;
  }
  BB_LOG(INFO) << "coordination reconnected to " << endpoints_[ep_cursor_]
               << " (" << watches_.size() << " watches restored)";
  return {};
}

Result<std::string> CoordClient::call_with_retry(uint16_t m, const std::string& body) {
  auto r = rpc_.call_raw(m, body);
  if (r.ok()) return r;
  // NOT_LEADER: we are talking to a read-only standby — advance to the next
  // endpoint (tried at most once per configured endpoint)
  size_t hops;
  {
    std::lock_guard<std::mutex> g(mu_);
    hops = endpoints_.size();
  }
  for (size_t attempt = 0; attempt < hops; ++attempt) {
    bool standby = r.code() == ErrorCode::NOT_LEADER;
    if (standby) {
      // NOT_LEADER messages carry the server's view of the fencing epoch as
      // "(epoch=N)" — track the max so a revived stale primary gets fenced
      auto pos = r.message().find("epoch=");
      if (pos != std::string::npos)
        observe_epoch(strtoull(r.message().c_str() + pos + 6, nullptr, 10));
    }
    switch (r.code()) {
      case ErrorCode::NOT_CONNECTED:
      case ErrorCode::CONNECTION_CLOSED:
      case ErrorCode::SEND_FAILED:
      case ErrorCode::RECV_FAILED:
      case ErrorCode::NOT_LEADER:
        break;
      default:
        return r;  // server-side error: no retry
    }
    std::function<void()> hook;
    {
      std::lock_guard<std::mutex> g(mu_);
      if (standby && endpoints_.size() < 2) return r;  // nowhere to go
      if (standby || !rpc_.connected()) {
        if (standby) ep_cursor_ = (ep_cursor_ + 1) % endpoints_.size();
        auto rd = redial_locked();
        if (!rd.ok()) return rd.error();
        hook = on_reconnect_;
      }
    }
    if (hook) hook();
    r = rpc_.call_raw(m, body);
    if (r.ok()) return r;
  }
  return r;
}

Result<void> CoordClient::put(const std::string& k, const std::string& v, uint64_t ttl) {
  auto r = call_with_retry(method::PUT, fenced(serde::to_bytes(PutReq{k, v, ttl})));
  if (!r.ok()) return r.error();
  return {};
}

Result<void> CoordClient::put_many(const std::vector<KV>& puts,
                                   const std::vector<std::string>& dels) {
  if (puts.empty() && dels.empty()) return {};
  auto r = call_with_retry(method::PUT_MANY,
                           fenced(serde::to_bytes(PutManyReq{puts, dels})));
  if (!r.ok()) return r.error();
  return {};
}

Result<std::string> CoordClient::get(const std::string& k) {
  auto r = call_with_retry(method::GET, serde::to_bytes(KeyReq{k}));
  if (!r.ok()) return r.error();
  ValueResp resp;
  if (!serde::from_bytes(r.value(), resp))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad GET response"};
  return resp.value;
}

Result<void> CoordClient::del(const std::string& k) {
  auto r = call_with_retry(method::DEL, fenced(serde::to_bytes(KeyReq{k})));
  if (!r.ok()) return r.error();
  return {};
}

Result<std::vector<KV>> CoordClient::get_prefix(const std::string& p) {
  auto r = call_with_retry(method::GET_PREFIX, serde::to_bytes(KeyReq{p}));
  if (!r.ok()) return r.error();
  PrefixResp resp;
  if (!serde::from_bytes(r.value(), resp))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad GET_PREFIX response"};
  return std::move(resp.kvs);
}

Result<bool> CoordClient::cas(const std::string& k, const std::string& e, bool ea,
                              const std::string& v, uint64_t ttl) {
  auto r = call_with_retry(
      method::CAS,
      fenced(serde::to_bytes(CasReq{k, e, v, static_cast<uint8_t>(ea), ttl})));
  if (!r.ok()) return r.error();
  BoolResp resp;
  if (!serde::from_bytes(r.value(), resp))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad CAS response"};
  return resp.ok != 0;
}

Result<void> CoordClient::keep_alive(const std::string& k, uint64_t ttl) {
  auto r = call_with_retry(method::KEEPALIVE,
                           fenced(serde::to_bytes(KeepAliveReq{k, ttl})));
  if (!r.ok()) return r.error();
  return {};
}

Result<uint64_t> CoordClient::watch_prefix(const std::string& p, WatchCallback cb) {
  auto r = call_with_retry(method::WATCH, serde::to_bytes(WatchReq{p}));
  if (!r.ok()) return r.error();
  WatchResp resp;
  if (!serde::from_bytes(r.value(), resp))
    return Error{ErrorCode::PROTOCOL_ERROR, "bad WATCH response"};
  std::lock_guard<std::mutex> g(mu_);
  uint64_t cid = next_client_watch_++;
  watches_[cid] = WatchEntry{p, std::move(cb), resp.watch_id};
  server_to_client_[resp.watch_id] = cid;
  return cid;
}

Result<void> CoordClient::unwatch(uint64_t cid) {
  uint64_t server_id = 0;
  {
    std::lock_guard<std::mutex> g(mu_);
    auto it = watches_.find(cid);
    if (it == watches_.end()) return {};
    server_id = it->second.server_id;
    server_to_client_.erase(server_id);
    watches_.erase(it);
  }
  auto r = call_with_retry(method::UNWATCH, serde::to_bytes(UnwatchReq{server_id}));
  if (!r.ok()) return r.error();
  return {};
}

std::shared_ptr<CoordService> make_coord(const std::string& endpoint) {
  if (endpoint.empty())
    return std::make_shared<InProcCoord>(std::make_shared<CoordStore>());
  auto c = std::make_shared<CoordClient>();
  auto r = c->connect(endpoint);
  if (!r.ok()) {
    BB_LOG(ERROR) << "coord connect failed: " << r.message();
    return nullptr;
  }
  return c;
}

// -------------------------------------------------------- LeaderElector

LeaderElector::LeaderElector(std::shared_ptr<CoordService> coord,
                             std::string election_key, std::string candidate_id,
                             uint64_t lease_ms)
    : coord_(std::move(coord)),
      key_(std::move(election_key)),
      id_(std::move(candidate_id)),
      lease_ms_(lease_ms) {}

LeaderElector::~LeaderElector() { stop(); }

void LeaderElector::start() {
  if (running_.exchange(true)) return;
  thread_ = std::thread([this] { loop(); });
}

void LeaderElector::stop() {
  if (!running_.exchange(false)) return;
  cv_.notify_all();
  if (thread_.joinable()) thread_.join();
  if (leader_.exchange(false)) {
    // resign: delete only if still ours
    auto cur = coord_->get(key_);
    if (cur.ok() && cur.value() == id_) coord_->del(key_);
  }
}

std::string LeaderElector::current_leader() {
  auto r = coord_->get(key_);
  return r.ok() ? r.value() : std::string{};
}

void LeaderElector::loop() {
  while (running_) {
    if (!leader_) {
      auto won = coord_->cas(key_, "", true, id_, lease_ms_);
      bool elected = false;
      if (won.ok() && won.value()) {
        elected = true;
        BB_LOG(INFO) << "leader elected: " << id_ << " on " << key_;
      } else {
        // maybe the old leader was us (restart) — steal our own key
        auto cur = coord_->get(key_);
        if (cur.ok() && cur.value() == id_) elected = true;
      }
      if (elected) {
        // run the promotion hook BEFORE advertising leadership: a promoted
        // keystone rescans persisted state in the hook, and requests that
        // slip in between the flag flip and the rescan would see an empty
        // object map (clients got NOT_FOUND for objects that exist)
        if (on_elected_) on_elected_();
        leader_ = true;
      }
    } else {
      auto r = coord_->keep_alive(key_, lease_ms_);
      if (!r.ok()) {
        // lease lost — re-campaign
        leader_ = false;
        BB_LOG(WARN) << "leadership lost: " << id_;
      }
    }
    std::unique_lock<std::mutex> lk(cv_mu_);
    cv_.wait_for(lk, std::chrono::milliseconds(lease_ms_ / 3 + 1),
                 [this] { return !running_.load(); });
  }
}

// ------------------------------------------------------- replication

CoordFollower::CoordFollower(std::shared_ptr<CoordStore> store,
                             CoordServer* server, std::string primary_endpoint,
                             uint64_t failover_ms)
    : store_(std::move(store)), server_(server),
      primary_(std::move(primary_endpoint)), failover_ms_(failover_ms) {}

CoordFollower::~CoordFollower() { stop(); }

Result<void> CoordFollower::sync_once() {
  BB_RETURN_IF_ERROR(rpc_.connect(primary_));
  // subscribe BEFORE the dump so nothing falls between them (events that
  // arrive during the dump re-apply idempotently)
  rpc_.set_event_callback([this](uint64_t, const std::string& body) {
    WatchEvent ev{};
    if (!serde::from_bytes(body, ev)) return;
    if (ev.type == EventType::PUT)
      (void)store_->put(ev.key, ev.value, ev.ttl_ms);
    else
      (void)store_->del(ev.key);
  });
  auto w = rpc_.call_raw(method::WATCH, serde::to_bytes(WatchReq{std::string{}}));
  if (!w.ok()) return w.error();
  auto d = rpc_.call_raw(method::DUMP, {});
  if (!d.ok()) return d.error();
  serde::Dec dec(d.value().data(), d.value().size());
  const uint64_t remote_epoch = dec.num<uint64_t>();
  std::vector<DumpEntry> entries;
  serde::get(dec, entries);
  if (!dec.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad DUMP"};
  // never re-follow a DEPOSED primary: if we promoted (or synced from a
  // newer leadership) our epoch is higher — mirroring an older generation
  // would resurrect the split brain the fencing epoch exists to prevent
  if (remote_epoch < store_->epoch())
    return Error{ErrorCode::NOT_LEADER,
                 "primary " + primary_ + " carries stale epoch " +
                     std::to_string(remote_epoch) + " < " +
                     std::to_string(store_->epoch())};
  store_->set_epoch(remote_epoch);
  for (auto& e : entries) (void)store_->put(e.key, e.value, e.ttl_ms);
  BB_LOG(INFO) << "coordd follower synced " << entries.size()
               << " keys from " << primary_;
  return {};
}

Result<void> CoordFollower::start() {
  if (running_.exchange(true)) return {};
  if (server_) server_->set_read_only(true);
  auto r = sync_once();
  if (!r.ok()) {
    running_ = false;
    if (server_) server_->set_read_only(false);
    return r;
  }
  monitor_ = std::thread([this] { monitor_loop(); });
  return {};
}

void CoordFollower::monitor_loop() {
  uint64_t down_since = 0;
  while (running_) {
    {
      std::unique_lock<std::mutex> lk(cv_mu_);
      cv_.wait_for(lk, std::chrono::milliseconds(100),
                   [this] { return !running_.load(); });
    }
    if (!running_) break;
    if (rpc_.connected()) {
      down_since = 0;
      continue;
    }
    const uint64_t now = now_ms();
    if (down_since == 0) {
      down_since = now;
      BB_LOG(WARN) << "coordd follower lost primary " << primary_;
    }
    // brief outage: try to resync (full DUMP recovers missed events;
    // deletions missed while down are not replayed — TTL'd state expires
    // on its own, which covers the liveness keys that matter)
    if (sync_once().ok()) {
      down_since = 0;
      continue;
    }
    if (now - down_since >= failover_ms_) {
      // bump the fencing epoch FIRST: every mutation accepted from here on
      // carries the new generation, so a revived old primary self-fences
      // the moment any failed-over client touches it
      const uint64_t e = store_->bump_epoch();
      BB_LOG(WARN) << "coordd follower PROMOTING after "
                   << (now - down_since) << " ms without a primary (epoch "
                   << e << ")";
      promoted_ = true;
      if (server_) server_->set_read_only(false);
      running_ = false;
      break;
    }
  }
}

void CoordFollower::stop() {
  running_ = false;
  {
    std::lock_guard<std::mutex> g(cv_mu_);  // no lost wakeup on stop
  }
  cv_.notify_all();
  if (monitor_.joinable()) monitor_.join();
  rpc_.close();
}

}  // namespace blackbird::coord
