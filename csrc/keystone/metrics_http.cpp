#include "blackbird/keystone/metrics_http.h"

#include <string.h>
#include <sys/socket.h>
#include <unistd.h>

#include <sstream>

#include "blackbird/common/log.h"
#include "blackbird/keystone/keystone_service.h"
#include "blackbird/rpc/net.h"

namespace blackbird {

MetricsHttpServer::MetricsHttpServer(KeystoneService& ks) : ks_(ks) {}

MetricsHttpServer::~MetricsHttpServer() { stop(); }

Result<void> MetricsHttpServer::start(const std::string& address) {
  auto hp = net::split_endpoint(address);
  if (!hp.ok()) return hp.error();
  uint16_t bound = 0;
  auto fd = net::listen_tcp(hp.value().first, hp.value().second, &bound);
  if (!fd.ok()) return fd.error();
  listen_fd_ = fd.value();
  port_ = bound;
  running_ = true;
  thread_ = std::thread([this] { serve_loop(); });
  BB_LOG(INFO) << "metrics http on port " << port_;
  return {};
}

void MetricsHttpServer::stop() {
  if (!running_.exchange(false)) return;
  ::shutdown(listen_fd_, SHUT_RDWR);
  ::close(listen_fd_);
  if (thread_.joinable()) thread_.join();
}

void MetricsHttpServer::serve_loop() {
  while (running_) {
    int cfd = ::accept(listen_fd_, nullptr, nullptr);
    if (cfd < 0) {
      if (!running_) break;
      continue;
    }
    // one-shot request/response; headers up to 4 KiB
    char buf[4096];
    ssize_t n = ::recv(cfd, buf, sizeof(buf) - 1, 0);
    if (n <= 0) {
      ::close(cfd);
      continue;
    }
    buf[n] = 0;
    std::string req(buf);
    std::string path = "/";
    auto sp1 = req.find(' ');
    auto sp2 = req.find(' ', sp1 + 1);
    if (sp1 != std::string::npos && sp2 != std::string::npos)
      path = req.substr(sp1 + 1, sp2 - sp1 - 1);

    std::string body, ctype = "text/plain; charset=utf-8";
    int code = 200;
    if (path == "/metrics") {
      body = render_metrics();
    } else if (path == "/stats") {
      body = render_stats();
      ctype = "application/json";
    } else if (path == "/healthz") {
      body = "ok\n";
    } else {
      code = 404;
      body = "not found\n";
    }
    std::ostringstream resp;
    resp << "HTTP/1.1 " << code << (code == 200 ? " OK" : " Not Found")
         << "\r\nContent-Type: " << ctype
         << "\r\nContent-Length: " << body.size()
         << "\r\nConnection: close\r\n\r\n"
         << body;
    auto s = resp.str();
    net::write_all(cfd, s.data(), s.size());
    ::close(cfd);
  }
}

std::string MetricsHttpServer::render_metrics() {
  auto st = ks_.get_cluster_stats();
  std::ostringstream os;
  os << "# HELP blackbird_capacity_bytes Total registered pool capacity\n"
     << "# TYPE blackbird_capacity_bytes gauge\n"
     << "blackbird_capacity_bytes " << st.total_capacity << "\n"
     << "# HELP blackbird_used_bytes Bytes reserved across all pools\n"
     << "# TYPE blackbird_used_bytes gauge\n"
     << "blackbird_used_bytes " << st.total_used << "\n"
     << "# TYPE blackbird_objects gauge\n"
     << "blackbird_objects " << st.num_objects << "\n"
     << "# TYPE blackbird_workers gauge\n"
     << "blackbird_workers " << st.num_workers << "\n"
     << "# TYPE blackbird_pools gauge\n"
     << "blackbird_pools " << st.num_pools << "\n"
     << "# TYPE blackbird_view_version counter\n"
     << "blackbird_view_version " << st.view_version << "\n";
  auto mc = ks_.counters();
  os << "# TYPE blackbird_migrations_total counter\n"
     << "blackbird_migrations_total " << mc.migrations << "\n"
     << "# TYPE blackbird_repairs_total counter\n"
     << "blackbird_repairs_total " << mc.repairs << "\n"
     << "# TYPE blackbird_scrub_quarantined_total counter\n"
     << "blackbird_scrub_quarantined_total " << mc.scrub_quarantined << "\n"
     << "# TYPE blackbird_evictions_total counter\n"
     << "blackbird_evictions_total " << mc.evictions << "\n"
     << "# TYPE blackbird_gc_reclaimed_total counter\n"
     << "blackbird_gc_reclaimed_total " << mc.gc_reclaimed << "\n"
     << "# HELP blackbird_token_commits_total Batch commits by session "
        "token (key-free fast-path commits)\n"
     << "# TYPE blackbird_token_commits_total counter\n"
     << "blackbird_token_commits_total " << ks_.token_commits() << "\n"
     << "# TYPE blackbird_is_leader gauge\n"
     << "blackbird_is_leader " << (ks_.is_leader() ? 1 : 0) << "\n";
  // per-pool gauges
  for (const auto& p : ks_.get_memory_pools()) {
    os << "blackbird_pool_used_bytes{pool=\"" << p.pool_id << "\",worker=\""
       << p.worker_id << "\",class=\"" << to_string(p.storage_class) << "\"} "
       << p.used << "\n";
    os << "blackbird_pool_capacity_bytes{pool=\"" << p.pool_id << "\",worker=\""
       << p.worker_id << "\",class=\"" << to_string(p.storage_class) << "\"} "
       << p.size << "\n";
  }
  return os.str();
}

std::string MetricsHttpServer::render_stats() {
  auto st = ks_.get_cluster_stats();
  json::Value v;
  v["total_capacity"] = st.total_capacity;
  v["total_used"] = st.total_used;
  v["num_objects"] = st.num_objects;
  v["num_workers"] = st.num_workers;
  v["num_pools"] = st.num_pools;
  v["view_version"] = st.view_version;
  v["is_leader"] = ks_.is_leader();
  json::Array pools;
  for (const auto& p : ks_.get_memory_pools()) pools.push_back(p.to_json());
  v["pools"] = std::move(pools);
  return v.dump();
}

}  // namespace blackbird
