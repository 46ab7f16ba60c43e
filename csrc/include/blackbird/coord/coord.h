// Coordination service — the discovery/liveness substrate.
// Capability parity with reference EtcdService (etcd_service.h:68-232): KV
// get/put/put-with-TTL/del, prefix scans, prefix watches, TTL leases,
// service registration, leader election. The reference shelled out to an
// external etcd cluster and left election/watch_key as stubs
// (etcd_service.cpp:295-298,379-385); this framework is self-contained: one
// CoordStore engine runs either embedded in-process (single-binary clusters,
// unit tests) or behind the framework's RPC framing as the `coordd` daemon —
// and leader election actually works (lease-protected compare-and-swap).
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "blackbird/common/result.h"
#include "blackbird/common/serde.h"
#include "blackbird/rpc/rpc.h"

namespace blackbird::coord {

enum class EventType : uint8_t { PUT = 0, DELETE = 1, EXPIRE = 2 };

struct WatchEvent {
  EventType type;
  std::string key;
  std::string value;   // empty for DELETE/EXPIRE
  uint64_t ttl_ms = 0; // remaining TTL at event time (0 = none) — lets a
                       // replication follower mirror leases faithfully
  BB_FIELDS(type, key, value, ttl_ms)
};

// One entry of a full-state DUMP (follower bootstrap / resync).
struct DumpEntry {
  std::string key;
  std::string value;
  uint64_t ttl_ms = 0;  // remaining
  BB_FIELDS(key, value, ttl_ms)
};

struct KV {
  std::string key;
  std::string value;
  BB_FIELDS(key, value)
};

using WatchCallback = std::function<void(const WatchEvent&)>;

// Abstract interface both the embedded and the remote client implement.
class CoordService {
 public:
  virtual ~CoordService() = default;
  virtual Result<void> put(const std::string& key, const std::string& value,
                           uint64_t ttl_ms = 0) = 0;
  virtual Result<std::string> get(const std::string& key) = 0;
  virtual Result<void> del(const std::string& key) = 0;
  virtual Result<std::vector<KV>> get_prefix(const std::string& prefix) = 0;
  // Compare-and-swap: succeeds iff current value == expected (expected empty
  // string + expect_absent=true means "key must not exist").
  virtual Result<bool> cas(const std::string& key, const std::string& expected,
                          bool expect_absent, const std::string& value,
                          uint64_t ttl_ms = 0) = 0;
  // Refresh a TTL'd key's lease without rewriting the value.
  virtual Result<void> keep_alive(const std::string& key, uint64_t ttl_ms) = 0;
  virtual Result<uint64_t> watch_prefix(const std::string& prefix, WatchCallback cb) = 0;
  virtual Result<void> unwatch(uint64_t watch_id) = 0;
  // batched puts (ttl 0) + deletes; remote implementations send ONE round
  // trip (synchronous durability of a whole commit batch — the role etcd
  // txn batches play for the reference's metadata, etcd_service.cpp:60-86)
  virtual Result<void> put_many(const std::vector<KV>& puts,
                                const std::vector<std::string>& dels) {
    for (const auto& kv : puts) BB_RETURN_IF_ERROR(put(kv.key, kv.value, 0));
    for (const auto& k : dels) BB_RETURN_IF_ERROR(del(k));
    return {};
  }
};

// ------------------------------------------------------------- the engine
class CoordStore {
 public:
  CoordStore();
  ~CoordStore();

  Result<void> put(const std::string& key, const std::string& value, uint64_t ttl_ms);
  Result<std::string> get(const std::string& key);
  Result<void> del(const std::string& key);
  Result<std::vector<KV>> get_prefix(const std::string& prefix);
  Result<bool> cas(const std::string& key, const std::string& expected,
                   bool expect_absent, const std::string& value, uint64_t ttl_ms);
  Result<void> keep_alive(const std::string& key, uint64_t ttl_ms);
  uint64_t add_watch(const std::string& prefix, WatchCallback cb);
  void remove_watch(uint64_t id);
  // Force one expiry sweep now (tests).
  void sweep_now();
  size_t size();

  // Durability: snapshot the KV map (values + absolute TTL deadlines) to a
  // file and restore it on restart — the role etcd's WAL played for the
  // reference. Atomic write (tmp + rename); expired entries dropped on load;
  // watches are session state and not persisted.
  Result<void> save(const std::string& path);
  Result<void> load(const std::string& path);
  // True if a put/del/cas/keep_alive happened since the last save().
  bool dirty() const { return dirty_.load(); }

  // ---- leadership fencing epoch ----
  // Monotonic leadership generation. A follower bumps it when it PROMOTES;
  // clients carry it on every mutation; a server that sees a NEWER epoch
  // than its own has been superseded and permanently demotes itself to
  // read-only (the revived-old-primary split-brain window of async
  // replication — etcd got this from raft terms; this is the fencing-token
  // equivalent). Persisted in snapshots, transferred in DUMP.
  uint64_t epoch() const { return epoch_.load(); }
  void set_epoch(uint64_t e) {
    epoch_.store(e);
    dirty_ = true;
  }
  uint64_t bump_epoch() {
    dirty_ = true;
    return epoch_.fetch_add(1) + 1;
  }
  // Full state with remaining TTLs (follower bootstrap; expired entries
  // skipped).
  std::vector<DumpEntry> dump();

 private:
  struct Entry {
    std::string value;
    uint64_t deadline_ms = 0;  // 0 = no TTL
  };
  struct Watch {
    std::string prefix;
    WatchCallback cb;
  };

  void sweeper_loop();
  void notify(EventType t, const std::string& key, const std::string& value,
              uint64_t ttl_ms = 0);

  std::mutex mu_;
  std::map<std::string, Entry> kv_;
  std::map<uint64_t, Watch> watches_;
  std::atomic<uint64_t> epoch_{1};
  std::atomic<bool> dirty_{false};
  std::atomic<uint64_t> next_watch_{1};
  std::atomic<bool> running_{true};
  std::condition_variable sweep_cv_;
  std::mutex sweep_mu_;
  std::thread sweeper_;
};

// ------------------------------------------- embedded (in-process) client
class InProcCoord : public CoordService {
 public:
  explicit InProcCoord(std::shared_ptr<CoordStore> store) : store_(std::move(store)) {}
  Result<void> put(const std::string& k, const std::string& v, uint64_t ttl) override {
    return store_->put(k, v, ttl);
  }
  Result<std::string> get(const std::string& k) override { return store_->get(k); }
  Result<void> del(const std::string& k) override { return store_->del(k); }
  Result<std::vector<KV>> get_prefix(const std::string& p) override {
    return store_->get_prefix(p);
  }
  Result<bool> cas(const std::string& k, const std::string& e, bool ea,
                   const std::string& v, uint64_t ttl) override {
    return store_->cas(k, e, ea, v, ttl);
  }
  Result<void> keep_alive(const std::string& k, uint64_t ttl) override {
    return store_->keep_alive(k, ttl);
  }
  Result<uint64_t> watch_prefix(const std::string& p, WatchCallback cb) override {
    return store_->add_watch(p, std::move(cb));
  }
  Result<void> unwatch(uint64_t id) override {
    store_->remove_watch(id);
    return {};
  }
  std::shared_ptr<CoordStore> store() { return store_; }

 private:
  std::shared_ptr<CoordStore> store_;
};

// ------------------------------------------------------------ TCP server
// Method ids 100-110 on the shared RPC framing.
namespace method {
constexpr uint16_t PUT = 100;
constexpr uint16_t GET = 101;
constexpr uint16_t DEL = 102;
constexpr uint16_t GET_PREFIX = 103;
constexpr uint16_t CAS = 104;
constexpr uint16_t KEEPALIVE = 105;
constexpr uint16_t WATCH = 106;
constexpr uint16_t UNWATCH = 107;
constexpr uint16_t DUMP = 108;  // full state incl. TTLs + epoch (replication)
constexpr uint16_t EPOCH = 109;  // current fencing epoch (u64)
constexpr uint16_t PUT_MANY = 110;  // batched puts + deletes, one round trip
}  // namespace method

class CoordServer {
 public:
  explicit CoordServer(std::shared_ptr<CoordStore> store);
  ~CoordServer();
  Result<void> start(const std::string& host, uint16_t port);
  void stop();
  uint16_t port() const { return rpc_.port(); }
  std::string endpoint() const { return rpc_.endpoint(); }
  std::shared_ptr<CoordStore> store() { return store_; }
  // Standby mode: mutations are rejected with NOT_LEADER (reads + watches
  // still served); a promoting follower flips this off.
  void set_read_only(bool ro) { read_only_.store(ro); }
  bool read_only() const { return read_only_.load(); }

 private:
  // write-barrier: read_only + fencing epoch (see CoordStore::epoch)
  Result<void> check_writable(uint64_t client_epoch);

  std::shared_ptr<CoordStore> store_;
  rpc::RpcServer rpc_;
  std::atomic<bool> read_only_{false};
  std::mutex mu_;
  std::map<uint64_t, std::vector<uint64_t>> conn_watches_;  // conn → watch ids
};

// ------------------------------------------------------- replication
// Async primary→standby mirroring: the follower full-syncs via DUMP, then
// applies the primary's watch events ("" prefix) to its local store. If the
// primary stays unreachable past failover_ms the follower PROMOTES: its
// server starts accepting writes and clients fail over to it (CoordClient
// accepts "epA,epB" endpoint lists). Replication is asynchronous — writes
// that the primary acknowledged but had not yet streamed are lost on
// failover (the reference deployed an external etcd for stronger
// guarantees; this covers single-node-loss continuity).
class CoordFollower {
 public:
  CoordFollower(std::shared_ptr<CoordStore> store, CoordServer* server,
                std::string primary_endpoint, uint64_t failover_ms = 2000);
  ~CoordFollower();
  Result<void> start();
  void stop();
  bool promoted() const { return promoted_.load(); }

 private:
  Result<void> sync_once();  // WATCH "" + DUMP apply
  void monitor_loop();

  std::shared_ptr<CoordStore> store_;
  CoordServer* server_;
  std::string primary_;
  uint64_t failover_ms_;
  rpc::RpcClient rpc_;
  std::atomic<bool> running_{false};
  std::atomic<bool> promoted_{false};
  std::thread monitor_;
  std::condition_variable cv_;
  std::mutex cv_mu_;
};

// ------------------------------------------------------------ TCP client
// Auto-reconnects after a coordination outage: calls that hit a dead
// connection retry once after redial, watches are re-subscribed on the new
// connection, and an optional on_reconnect hook lets services re-register
// state the (in-memory) coordination server lost.
class CoordClient : public CoordService {
 public:
  CoordClient() = default;
  ~CoordClient() override;
  // endpoint may be a comma-separated list ("epA,epB"): the client connects
  // to the first reachable one and cycles through the list on connection
  // loss or NOT_LEADER (standby) responses — coordd failover is transparent
  Result<void> connect(const std::string& endpoint, int timeout_ms = 5000);
  void close();
  // invoked (on the calling thread) after a successful reconnect
  void set_on_reconnect(std::function<void()> cb);

  Result<void> put(const std::string& k, const std::string& v, uint64_t ttl) override;
  Result<std::string> get(const std::string& k) override;
  Result<void> del(const std::string& k) override;
  Result<std::vector<KV>> get_prefix(const std::string& p) override;
  Result<bool> cas(const std::string& k, const std::string& e, bool ea,
                   const std::string& v, uint64_t ttl) override;
  Result<void> keep_alive(const std::string& k, uint64_t ttl) override;
  Result<uint64_t> watch_prefix(const std::string& p, WatchCallback cb) override;
  Result<void> unwatch(uint64_t id) override;
  Result<void> put_many(const std::vector<KV>& puts,
                        const std::vector<std::string>& dels) override;
  // highest leadership epoch observed (sent with every mutation; a stale
  // revived primary rejects + self-fences on seeing a newer one)
  uint64_t observed_epoch() const { return epoch_.load(); }

 private:
  // retry wrapper: redial + re-subscribe watches on connection loss
  Result<std::string> call_with_retry(uint16_t method, const std::string& body);
  // mutation body + trailing fencing epoch
  std::string fenced(std::string body);
  void refresh_epoch_locked();
  void observe_epoch(uint64_t e);
  Result<void> redial_locked();
  void install_event_callback();

  std::vector<std::string> endpoints_;
  size_t ep_cursor_ = 0;
  int timeout_ms_ = 5000;
  rpc::RpcClient rpc_;
  std::mutex mu_;            // guards watch tables + redial
  struct WatchEntry {
    std::string prefix;
    WatchCallback cb;
    uint64_t server_id = 0;  // id on the CURRENT connection
  };
  std::map<uint64_t, WatchEntry> watches_;   // client-stable id → entry
  std::map<uint64_t, uint64_t> server_to_client_;  // server id → client id
  uint64_t next_client_watch_ = 1;
  std::function<void()> on_reconnect_;
  std::atomic<uint64_t> epoch_{0};  // max leadership epoch seen
};

// Build the right client for an endpoint ("" = fresh embedded store).
std::shared_ptr<CoordService> make_coord(const std::string& endpoint);

// ------------------------------------------------------- leader election
// CAS a lease-protected key; the holder refreshes it, others watch and
// re-campaign on expiry. (The reference's campaign_leader was a stub.)
class LeaderElector {
 public:
  LeaderElector(std::shared_ptr<CoordService> coord, std::string election_key,
                std::string candidate_id, uint64_t lease_ms = 5000);
  ~LeaderElector();
  void start();
  void stop();
  bool is_leader() const { return leader_.load(); }
  std::string current_leader();
  // invoked (on the elector thread) each time this candidate BECOMES the
  // leader — a promoted standby uses it to rescan persisted cluster state
  void set_on_elected(std::function<void()> cb) { on_elected_ = std::move(cb); }

 private:
  void loop();
  std::shared_ptr<CoordService> coord_;
  std::string key_;
  std::string id_;
  uint64_t lease_ms_;
  std::atomic<bool> leader_{false};
  std::atomic<bool> running_{false};
  std::function<void()> on_elected_;
  std::thread thread_;
  std::condition_variable cv_;
  std::mutex cv_mu_;
};

}  // namespace blackbird::coord
