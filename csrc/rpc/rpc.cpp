#include "blackbird/rpc/rpc.h"

#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include "blackbird/common/log.h"
#include "blackbird/rpc/net.h"

namespace blackbird::rpc {

#pragma pack(push, 1)
struct WireHeader {
  uint32_t body_len;
  uint8_t kind;
  uint64_t id;
  uint16_t method;
};
#pragma pack(pop)
static_assert(sizeof(WireHeader) == 15);

Result<void> send_frame(int fd, FrameKind kind, uint64_t id, uint16_t method,
                        const void* body, size_t body_len, std::mutex* write_mu) {
  if (body_len > kMaxFrame)
    return Error{ErrorCode::PROTOCOL_ERROR, "frame too large"};
  WireHeader h{static_cast<uint32_t>(body_len), static_cast<uint8_t>(kind), id, method};
  if (write_mu) {
    std::lock_guard<std::mutex> g(*write_mu);
    return net::write_all2(fd, &h, sizeof(h), body, body_len);
  }
  return net::write_all2(fd, &h, sizeof(h), body, body_len);
}

Result<Frame> recv_frame(int fd) {
  WireHeader h;
  BB_RETURN_IF_ERROR(net::read_exact(fd, &h, sizeof(h)));
  if (h.body_len > kMaxFrame)
    return Error{ErrorCode::PROTOCOL_ERROR, "oversized frame"};
  Frame f;
  f.kind = static_cast<FrameKind>(h.kind);
  f.id = h.id;
  f.method = h.method;
  f.body.resize(h.body_len);
  if (h.body_len > 0)
    BB_RETURN_IF_ERROR(net::read_exact(fd, f.body.data(), h.body_len));
  return f;
}

// ----------------------------------------------------------------- server

RpcServer::~RpcServer() { stop(); }

void RpcServer::register_handler(uint16_t method, Handler h) {
  handlers_[method] = std::move(h);
}

Result<void> RpcServer::start(const std::string& host, uint16_t port) {
  if (net::is_unix_endpoint(host)) {
    auto fd = net::listen_unix(host.substr(5));
    if (!fd.ok()) return fd.error();
    listen_fd_ = fd.value();
    port_ = 0;
    host_ = host;
    running_ = true;
    accept_thread_ = std::thread([this] { accept_loop(); });
    return {};
  }
  uint16_t bound = 0;
  auto fd = net::listen_tcp(host, port, &bound);
  if (!fd.ok()) return fd.error();
  listen_fd_ = fd.value();
  port_ = bound;
  host_ = host;
  running_ = true;
  accept_thread_ = std::thread([this] { accept_loop(); });
  return {};
}

std::string RpcServer::endpoint() const {
  if (net::is_unix_endpoint(host_)) return host_;
  std::string h = (host_ == "0.0.0.0" || host_.empty()) ? net::advertise_host() : host_;
  return h + ":" + std::to_string(port_);
}

void RpcServer::stop() {
  if (!running_.exchange(false)) return;
  int lfd = listen_fd_.exchange(-1);
  if (lfd >= 0) {
    ::shutdown(lfd, SHUT_RDWR);
    ::close(lfd);
  }
  if (accept_thread_.joinable()) accept_thread_.join();
  std::map<uint64_t, Conn> conns;
  {
    std::lock_guard<std::mutex> g(conns_mu_);
    conns.swap(conns_);
  }
  for (auto& [id, c] : conns) {
    ::shutdown(c.fd, SHUT_RDWR);
    if (c.thread.joinable()) c.thread.join();
    std::lock_guard<std::mutex> g(*c.write_mu);  // no in-flight event push
    c.alive->store(false);
    ::close(c.fd);
  }
}

void RpcServer::accept_loop() {
  while (running_) {
    int cfd = ::accept(listen_fd_.load(), nullptr, nullptr);
    if (cfd < 0) {
      if (!running_) break;
      continue;
    }
    net::set_nodelay(cfd);
    uint64_t id = next_conn_id_++;
    std::lock_guard<std::mutex> g(conns_mu_);
    auto& c = conns_[id];
    c.fd = cfd;
    c.write_mu = std::make_shared<std::mutex>();
    c.alive = std::make_shared<std::atomic<bool>>(true);
    c.thread = std::thread([this, cfd, id] { conn_loop(cfd, id); });
  }
}

void RpcServer::conn_loop(int fd, uint64_t conn_id) {
  std::shared_ptr<std::mutex> wmu;
  std::shared_ptr<std::atomic<bool>> alive;
  {
    std::lock_guard<std::mutex> g(conns_mu_);
    auto it = conns_.find(conn_id);
    if (it != conns_.end()) {
      wmu = it->second.write_mu;
      alive = it->second.alive;
    }
  }
  if (!wmu) return;

  ConnCtx ctx;
  ctx.conn_id = conn_id;
  ctx.push = [fd, wmu, alive](uint64_t event_id, const std::string& body) {
    // hold the write mutex across the liveness check AND the send: closers
    // take the same mutex before closing the fd
    std::lock_guard<std::mutex> g(*wmu);
    if (!alive->load()) return;
    send_frame(fd, FrameKind::EVENT, event_id, 0, body.data(), body.size(),
               nullptr);
  };

  while (running_) {
    auto fr = recv_frame(fd);
    if (!fr.ok()) break;
    Frame& f = fr.value();
    if (f.kind != FrameKind::REQUEST) continue;

    serde::Enc resp;
    auto it = handlers_.find(f.method);
    if (it == handlers_.end()) {
      resp.num<int32_t>(static_cast<int32_t>(ErrorCode::NOT_IMPLEMENTED));
      resp.str("unknown method " + std::to_string(f.method));
    } else {
      Result<std::string> r = ErrorCode::INTERNAL_ERROR;
      try {
        r = it->second(f.body, ctx);
      } catch (const std::exception& e) {
        r = Error{ErrorCode::INTERNAL_ERROR, e.what()};
      }
      if (r.ok()) {
        resp.num<int32_t>(0);
        resp.str("");
        resp.buf.append(r.value());
      } else {
        resp.num<int32_t>(static_cast<int32_t>(r.code()));
        resp.str(r.message());
      }
    }
    auto sr = send_frame(fd, FrameKind::RESPONSE, f.id, f.method, resp.buf.data(),
                         resp.buf.size(), wmu.get());
    if (!sr.ok()) break;
  }

  if (disconnect_cb_) disconnect_cb_(conn_id);
  // Detach our entry; let stop() or us close the fd exactly once.
  std::lock_guard<std::mutex> g(conns_mu_);
  auto it = conns_.find(conn_id);
  if (it != conns_.end()) {
    it->second.thread.detach();
    {
      std::lock_guard<std::mutex> wg(*it->second.write_mu);
      it->second.alive->store(false);
      ::close(it->second.fd);
    }
    conns_.erase(it);
  }
}

// ----------------------------------------------------------------- client

RpcClient::~RpcClient() { close(); }

Result<void> RpcClient::connect(const std::string& endpoint, int timeout_ms) {
  if (net::is_unix_endpoint(endpoint)) {
    close();
    auto fd = net::connect_unix(endpoint.substr(5), timeout_ms);
    if (!fd.ok()) return fd.error();
    fd_ = fd.value();
    running_ = true;
    reader_ = std::thread([this] { reader_loop(); });
    dispatcher_ = std::thread([this] { dispatch_loop(); });
    return {};
  }
  auto hp = net::split_endpoint(endpoint);
  if (!hp.ok()) return hp.error();
  return connect(hp.value().first, hp.value().second, timeout_ms);
}

Result<void> RpcClient::connect(const std::string& host, uint16_t port, int timeout_ms) {
  close();
  auto fd = net::connect_tcp(host, port, timeout_ms);
  if (!fd.ok()) return fd.error();
  fd_ = fd.value();
  running_ = true;
  reader_ = std::thread([this] { reader_loop(); });
  dispatcher_ = std::thread([this] { dispatch_loop(); });
  return {};
}

void RpcClient::close() {
  // Safe against concurrent close()/destructor races and against being
  // invoked from the dispatcher thread itself (an event callback dropping
  // the last reference): joins become detaches for the calling thread.
  std::lock_guard<std::mutex> cg(close_mu_);
  running_ = false;
  if (fd_ >= 0) ::shutdown(fd_, SHUT_RDWR);
  // notify WITH evq_mu_ held: the dispatcher's predicate reads running_, so
  // an unlocked notify can be lost between its predicate check and the
  // park (observed as a rare dispatcher-join hang under TSan)
  auto wake_dispatcher = [this] {
    std::lock_guard<std::mutex> g(evq_mu_);
    evq_cv_.notify_all();
  };
  wake_dispatcher();
  auto join_or_detach = [](std::thread& t) {
    if (!t.joinable()) return;
    if (t.get_id() == std::this_thread::get_id()) t.detach();
    else t.join();
  };
  // a reader that exited on its own (peer closed) is still joinable
  join_or_detach(reader_);
  wake_dispatcher();
  join_or_detach(dispatcher_);
  {
    // a user thread may still be inside call_raw's send (failover paths
    // reconnect while other lanes issue calls): the send holds write_mu_,
    // so closing under it can never hit a mid-sendmsg (or reused) fd
    std::lock_guard<std::mutex> g(write_mu_);
    if (fd_ >= 0) {
      ::close(fd_);
      fd_ = -1;
    }
  }
  fail_all_pending(ErrorCode::CONNECTION_CLOSED);
}

void RpcClient::set_event_callback(EventCallback cb) {
  std::lock_guard<std::mutex> g(event_cb_mu_);
  event_cb_ = std::move(cb);
}

void RpcClient::fail_all_pending(ErrorCode code) {
  std::lock_guard<std::mutex> g(mu_);
  for (auto& [id, p] : pending_) {
    p->failed = true;
    p->status = static_cast<int32_t>(code);
    p->done.store(true, std::memory_order_release);
  }
  cv_.notify_all();
}

void RpcClient::reader_loop() {
  while (running_) {
    auto fr = recv_frame(fd_);
    if (!fr.ok()) break;
    Frame& f = fr.value();
    if (f.kind == FrameKind::RESPONSE) {
      std::lock_guard<std::mutex> g(mu_);
      auto it = pending_.find(f.id);
      if (it != pending_.end()) {
        it->second->body = std::move(f.body);
        it->second->done.store(true, std::memory_order_release);
        cv_.notify_all();
      }
    } else if (f.kind == FrameKind::EVENT) {
      std::lock_guard<std::mutex> g(evq_mu_);
      evq_.emplace_back(f.id, std::move(f.body));
      evq_cv_.notify_one();
    }
  }
  running_ = false;
  fail_all_pending(ErrorCode::CONNECTION_CLOSED);
}


void RpcClient::dispatch_loop() {
  while (true) {
    std::vector<std::pair<uint64_t, std::string>> batch;
    {
      std::unique_lock<std::mutex> lk(evq_mu_);
      evq_cv_.wait(lk, [this] { return !evq_.empty() || !running_.load(); });
      if (evq_.empty() && !running_) break;
      batch.swap(evq_);
    }
    EventCallback cb;
    {
      std::lock_guard<std::mutex> g(event_cb_mu_);
      cb = event_cb_;
    }
    if (cb)
      for (auto& [id, body] : batch) cb(id, body);
  }
}

Result<std::string> RpcClient::call_raw(uint16_t method, const std::string& body,
                                        int timeout_ms) {
  if (fd_ < 0 || !running_) return Error{ErrorCode::NOT_CONNECTED, "not connected"};
  uint64_t id = next_req_++;
  auto p = std::make_shared<Pending>();
  {
    std::lock_guard<std::mutex> g(mu_);
    pending_[id] = p;
  }
  Result<void> sr{};
  {
    // fd_ read + send under ONE write_mu_ hold: serialized with close()
    std::lock_guard<std::mutex> g(write_mu_);
    if (fd_ < 0 || !running_)
      sr = Error{ErrorCode::NOT_CONNECTED, "not connected"};
    else
      sr = send_frame(fd_, FrameKind::REQUEST, id, method, body.data(),
                      body.size(), nullptr);
  }
  if (!sr.ok()) {
    std::lock_guard<std::mutex> g(mu_);
    pending_.erase(id);
    return sr.error();
  }
  // bounded spin before sleeping: the reply for a small metadata call
  // arrives in tens of µs, and a cv sleep on an idle core costs far more
  // to wake than the answer took to compute
  for (int spin = 0; spin < 4000 && !p->done.load(std::memory_order_acquire);
       ++spin) {
#if defined(__x86_64__)
    __builtin_ia32_pause();
#else
    std::this_thread::yield();
#endif
  }
  std::unique_lock<std::mutex> lk(mu_);
  bool ok = p->done.load(std::memory_order_acquire) ||
            cv_.wait_for(lk, std::chrono::milliseconds(timeout_ms),
                         [&] { return p->done.load(std::memory_order_acquire); });
  pending_.erase(id);
  if (!ok) return Error{ErrorCode::TIMEOUT, "rpc timeout (method " +
                                              std::to_string(method) + ")"};
  if (p->failed) return Error{static_cast<ErrorCode>(p->status), "connection lost"};
  lk.unlock();

  // decode [i32 status][str message][payload]
  serde::Dec d(p->body.data(), p->body.size());
  int32_t status = d.num<int32_t>();
  std::string msg = d.str();
  if (!d.ok()) return Error{ErrorCode::PROTOCOL_ERROR, "bad response frame"};
  if (status != 0) return Error{static_cast<ErrorCode>(status), std::move(msg)};
  return std::string(p->body.data() + (p->body.size() - d.remaining()), d.remaining());
}

}  // namespace blackbird::rpc
