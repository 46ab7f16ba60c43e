// Core data model: storage classes, pool/worker descriptors, placement
// structs, object metadata, RPC messages, configs.
// Capability parity with reference include/blackbird/common/types.h:82-493,
// re-designed for MI355X: pools carry an AccessInfo that generalizes the
// reference's {ucx_endpoint, ucx_remote_addr, ucx_rkey_hex} advertisement
// (types.h:464-493) to {tcp | shm | hip_ipc} access paths, with RAM_GPU a
// first-class HBM3E tier instead of the reference's stub
// (worker_service.cpp:196).
#pragma once

#include <chrono>
#include <cstdint>
#include <map>
#include <optional>
#include <string>
#include <vector>

#include "blackbird/common/json.h"
#include "blackbird/common/serde.h"

namespace blackbird {

using ObjectKey = std::string;
using WorkerId = std::string;
using PoolId = std::string;

inline uint64_t now_ms() {
  return static_cast<uint64_t>(
      std::chrono::duration_cast<std::chrono::milliseconds>(
          std::chrono::steady_clock::now().time_since_epoch())
          .count());
}
inline uint64_t wall_ms() {
  return static_cast<uint64_t>(
      std::chrono::duration_cast<std::chrono::milliseconds>(
          std::chrono::system_clock::now().time_since_epoch())
          .count());
}

// ---------------------------------------------------------------- storage
enum class StorageClass : uint8_t {
  RAM_CPU = 0,    // host DRAM (malloc or POSIX shm)
  RAM_GPU = 1,    // MI355X HBM3E — first-class tier
  PINNED_CPU = 2, // page-locked host memory (hipHostMalloc) — staging tier
  NVME = 3,
  SSD = 4,
  HDD = 5,
  CXL_MEM = 6,  // CXL.mem expander: DAX-device mmap (anonymous fallback)
};

inline const char* to_string(StorageClass c) {
  switch (c) {
    case StorageClass::RAM_CPU: return "RAM_CPU";
    case StorageClass::RAM_GPU: return "RAM_GPU";
    case StorageClass::PINNED_CPU: return "PINNED_CPU";
    case StorageClass::NVME: return "NVME";
    case StorageClass::SSD: return "SSD";
    case StorageClass::HDD: return "HDD";
    case StorageClass::CXL_MEM: return "CXL_MEM";
  }
  return "UNKNOWN";
}

inline std::optional<StorageClass> storage_class_from_string(const std::string& s) {
  if (s == "RAM_CPU") return StorageClass::RAM_CPU;
  if (s == "RAM_GPU") return StorageClass::RAM_GPU;
  if (s == "PINNED_CPU") return StorageClass::PINNED_CPU;
  if (s == "NVME") return StorageClass::NVME;
  if (s == "SSD") return StorageClass::SSD;
  if (s == "HDD") return StorageClass::HDD;
  if (s == "CXL_MEM") return StorageClass::CXL_MEM;
  return std::nullopt;
}

// Tier ordering for promotion/demotion (lower = faster).
inline int tier_rank(StorageClass c) {
  switch (c) {
    case StorageClass::RAM_GPU: return 0;
    case StorageClass::PINNED_CPU: return 1;
    case StorageClass::RAM_CPU: return 2;
    case StorageClass::CXL_MEM: return 3;  // DRAM-class latency, below local
    case StorageClass::NVME: return 4;
    case StorageClass::SSD: return 5;
    case StorageClass::HDD: return 6;
  }
  return 7;
}

// How a client reaches a pool's memory. Replaces the reference's UCX rkey
// advertisement (MemoryPool.ucx_* fields, types.h:474-476).
enum class AccessKind : uint8_t {
  TCP = 0,      // framed data protocol to the worker's data port (universal)
  SHM = 1,      // same-host POSIX shared memory — one-sided memcpy
  HIP_IPC = 2,  // same-node HBM via hipIpcMemHandle — one-sided hipMemcpy
};

struct AccessInfo {
  AccessKind kind = AccessKind::TCP;
  std::string endpoint;         // host:port of worker data plane (always set)
  std::string shm_name;         // SHM: /bb_pool_<id>
  int32_t device_id = -1;       // HIP_IPC: HIP device ordinal on the worker
  std::string ipc_handle_hex;   // HIP_IPC: hex hipIpcMemHandle_t
  uint64_t base_addr = 0;       // worker-side VA base (diagnostics)

  BB_FIELDS(kind, endpoint, shm_name, device_id, ipc_handle_hex, base_addr)

  json::Value to_json() const {
    json::Value v;
    v["kind"] = static_cast<int64_t>(kind);
    v["endpoint"] = endpoint;
    v["shm_name"] = shm_name;
    v["device_id"] = static_cast<int64_t>(device_id);
    v["ipc_handle_hex"] = ipc_handle_hex;
    v["base_addr"] = base_addr;
    return v;
  }
  static AccessInfo from_json(const json::Value& v) {
    AccessInfo a;
    a.kind = static_cast<AccessKind>(v["kind"].i64());
    a.endpoint = v["endpoint"].str();
    a.shm_name = v["shm_name"].str();
    a.device_id = static_cast<int32_t>(v["device_id"].i64(-1));
    a.ipc_handle_hex = v["ipc_handle_hex"].str();
    a.base_addr = v["base_addr"].u64();
    return a;
  }
};

// Cluster-visible pool descriptor (advertised via coordination as JSON).
// Parity: reference MemoryPool types.h:464-493.
struct MemoryPool {
  PoolId pool_id;
  WorkerId worker_id;
  std::string node_id;
  StorageClass storage_class = StorageClass::RAM_CPU;
  uint64_t size = 0;
  uint64_t used = 0;
  AccessInfo access;

  uint64_t available() const { return size > used ? size - used : 0; }

  BB_FIELDS(pool_id, worker_id, node_id, storage_class, size, used, access)

  json::Value to_json() const {
    json::Value v;
    v["pool_id"] = pool_id;
    v["worker_id"] = worker_id;
    v["node_id"] = node_id;
    v["storage_class"] = std::string(to_string(storage_class));
    v["size"] = size;
    v["used"] = used;
    v["access"] = access.to_json();
    return v;
  }
  static MemoryPool from_json(const json::Value& v) {
    MemoryPool p;
    p.pool_id = v["pool_id"].str();
    p.worker_id = v["worker_id"].str();
    p.node_id = v["node_id"].str();
    p.storage_class =
        storage_class_from_string(v["storage_class"].str()).value_or(StorageClass::RAM_CPU);
    p.size = v["size"].u64();
    p.used = v["used"].u64();
    p.access = AccessInfo::from_json(v["access"]);
    return p;
  }
};

// ------------------------------------------------------------- placement
// One contiguous shard of one copy, in one pool.
// Parity: reference ShardPlacement types.h:139-149 (denormalized with the
// pool's AccessInfo so clients never re-query pool metadata mid-transfer).
struct ShardPlacement {
  PoolId pool_id;
  WorkerId worker_id;
  StorageClass storage_class = StorageClass::RAM_CPU;
  uint64_t offset = 0;  // pool-relative byte offset
  uint64_t length = 0;
  AccessInfo access;
  // standalone bbhash64 of THIS shard's bytes (0 = not recorded). Recorded
  // at put time so the scrubber can verify striped copies shard by shard —
  // the whole-object digest is position-weighted and not combinable across
  // shard boundaries.
  uint64_t digest = 0;

  BB_FIELDS(pool_id, worker_id, storage_class, offset, length, access, digest)

  bool operator==(const ShardPlacement& o) const {
    return pool_id == o.pool_id && offset == o.offset && length == o.length;
  }
};

// One full replica of an object = ordered shards covering [0, size).
// Parity: reference CopyPlacement types.h:151-156.
struct CopyPlacement {
  uint32_t copy_index = 0;
  std::vector<ShardPlacement> shards;

  BB_FIELDS(copy_index, shards)
};

// Client-requested placement constraints.
// Parity: reference WorkerConfig types.h:161-187.
struct PlacementConfig {
  uint32_t replication = 1;          // number of copies
  uint32_t max_workers_per_copy = 1; // striping degree
  uint64_t min_shard_size = 4096;
  std::optional<StorageClass> preferred_class;
  // hard constraint (tier migration targets); preferred_class is a soft one
  std::optional<StorageClass> required_class;
  uint64_t ttl_ms = 0;               // 0 = no expiry
  bool checksum = true;              // compute/verify GPU checksum
  // soft locality hint: place copy 0 on this worker when it has room (a
  // rank keeps its working set in local HBM instead of scattering over
  // xGMI; later copies still spread for fault tolerance)
  WorkerId preferred_worker;
  // upsert: an existing committed object under the same key is atomically
  // removed and re-placed (put_start otherwise rejects with OBJECT_EXISTS).
  // Saves the separate remove RPC for key-cycling workloads.
  bool replace = false;

  BB_FIELDS(replication, max_workers_per_copy, min_shard_size, preferred_class,
            required_class, ttl_ms, checksum, preferred_worker, replace)
};

enum class ObjectState : uint8_t { PENDING = 0, COMMITTED = 1 };

struct ObjectMeta {
  ObjectKey key;
  uint64_t size = 0;
  uint64_t checksum = 0;  // 64-bit MFMA digest (0 = not computed)
  uint64_t ttl_ms = 0;
  uint64_t created_ms = 0;     // steady-clock ms
  uint64_t last_access_ms = 0;
  uint32_t access_count = 0;   // accesses since the last tiering cycle
  uint32_t replication = 1;    // desired copy count (repair target)
  uint64_t last_scrub_ms = 0;  // when the scrubber last verified the digests
  ObjectState state = ObjectState::PENDING;
  std::vector<CopyPlacement> copies;

  bool expired(uint64_t now) const {
    return ttl_ms > 0 && state == ObjectState::COMMITTED &&
           now > created_ms + ttl_ms;
  }

  BB_FIELDS(key, size, checksum, ttl_ms, created_ms, last_access_ms,
            access_count, replication, last_scrub_ms, state, copies)
};

// One row of a LIST_OBJECTS prefix scan (operator tooling: `bbctl ls`).
struct ObjectSummary {
  ObjectKey key;
  uint64_t size = 0;
  uint32_t ncopies = 0;
  StorageClass storage_class = StorageClass::RAM_CPU;  // of copy 0
  BB_FIELDS(key, size, ncopies, storage_class)
};

// ---------------------------------------------------------------- workers
struct WorkerInfo {
  WorkerId worker_id;
  std::string node_id;
  std::string data_endpoint;  // host:port of data plane
  uint64_t registered_ms = 0;
  uint64_t last_heartbeat_ms = 0;

  bool is_stale(uint64_t now, uint64_t ttl) const {
    return last_heartbeat_ms + ttl < now;
  }

  BB_FIELDS(worker_id, node_id, data_endpoint, registered_ms, last_heartbeat_ms)

  json::Value to_json() const {
    json::Value v;
    v["worker_id"] = worker_id;
    v["node_id"] = node_id;
    v["data_endpoint"] = data_endpoint;
    v["registered_ms"] = registered_ms;
    return v;
  }
  static WorkerInfo from_json(const json::Value& v) {
    WorkerInfo w;
    w.worker_id = v["worker_id"].str();
    w.node_id = v["node_id"].str();
    w.data_endpoint = v["data_endpoint"].str();
    w.registered_ms = v["registered_ms"].u64();
    return w;
  }
};

struct ClusterStats {
  uint64_t total_capacity = 0;
  uint64_t total_used = 0;
  uint64_t num_objects = 0;
  uint64_t num_workers = 0;
  uint64_t num_pools = 0;
  uint64_t view_version = 0;

  BB_FIELDS(total_capacity, total_used, num_objects, num_workers, num_pools,
            view_version)
};

// ---------------------------------------------------------------- configs
struct KeystoneConfig {
  std::string cluster_id = "default";
  std::string listen_address = "0.0.0.0:9090";
  std::string coord_endpoint;        // empty = embedded in-process coordination
  std::string metrics_address;       // host:port for /metrics; empty = off
  uint64_t object_ttl_default_ms = 0;
  uint64_t gc_interval_ms = 5000;
  uint64_t health_interval_ms = 2000;
  uint64_t worker_ttl_ms = 10000;
  double eviction_high_watermark = 0.85;  // start evicting above this fill
  double eviction_ratio = 0.10;           // evict this fraction of objects
  bool enable_ha = false;                 // leader election via coordination
  // ---- tier migration (GPU→DRAM→NVMe spill, promotion of hot objects) ----
  bool enable_tiering = true;
  double tier_high_watermark = 0.80;      // demote when a tier fills past this
  uint32_t tier_max_moves_per_cycle = 32;
  uint32_t promote_hot_threshold = 4;     // accesses/cycle; 0 = no promotion
  // auto-compact a pool when 1 - largest_free/total_free exceeds this
  // (0 = compaction only via the explicit compact_pool API)
  double compact_fragmentation_threshold = 0.0;
  // re-replication throttle: at most this many objects repaired per GC pass
  uint32_t repair_max_per_cycle = 16;
  // background digest scrubbing: every interval, re-checksum up to
  // scrub_batch single-shard copies on their workers and quarantine
  // mismatching copies (repair restores replication). 0 = off.
  uint64_t scrub_interval_ms = 0;
  uint32_t scrub_batch = 64;
  // persist object metadata to the coordination service so a keystone
  // restart keeps the object map (the reference lost it)
  bool persist_objects = false;

  BB_FIELDS(cluster_id, listen_address, coord_endpoint, metrics_address,
            object_ttl_default_ms, gc_interval_ms, health_interval_ms,
            worker_ttl_ms, eviction_high_watermark, eviction_ratio, enable_ha,
            enable_tiering, tier_high_watermark, tier_max_moves_per_cycle,
            promote_hot_threshold, compact_fragmentation_threshold,
            repair_max_per_cycle, scrub_interval_ms, scrub_batch,
            persist_objects)
};

struct PoolConfig {
  PoolId pool_id;
  StorageClass storage_class = StorageClass::RAM_CPU;
  uint64_t size_bytes = 0;
  std::string mount_path;   // disk tiers
  int32_t gpu_device_id = 0;

  BB_FIELDS(pool_id, storage_class, size_bytes, mount_path, gpu_device_id)
};

struct WorkerConfig {
  WorkerId worker_id;
  std::string node_id;
  std::string cluster_id = "default";
  std::string coord_endpoint;
  std::string data_listen_address = "0.0.0.0:0";  // port 0 = auto
  uint64_t heartbeat_interval_ms = 2000;
  uint64_t heartbeat_ttl_ms = 6000;
  std::vector<PoolConfig> pools;

  BB_FIELDS(worker_id, node_id, cluster_id, coord_endpoint, data_listen_address,
            heartbeat_interval_ms, heartbeat_ttl_ms, pools)
};

// ----------------------------------------------------------- RPC messages
// The 14-method surface of the reference RpcService (rpc_service.h:200-265)
// plus batch ops (types.h:333-392).

struct PutStartRequest {
  ObjectKey key;
  uint64_t size = 0;
  PlacementConfig config;
  BB_FIELDS(key, size, config)
};

struct PutStartResponse {
  std::vector<CopyPlacement> copies;
  uint64_t view_version = 0;
  BB_FIELDS(copies, view_version)
};

struct PutCompleteRequest {
  ObjectKey key;
  uint64_t checksum = 0;
  // per copy, per shard: standalone digests of the striped pieces (empty
  // for single-shard puts — the whole-object checksum covers those)
  std::vector<std::vector<uint64_t>> shard_digests;
  BB_FIELDS(key, checksum, shard_digests)
};

struct GetWorkersResponse {
  std::vector<CopyPlacement> copies;
  uint64_t size = 0;
  uint64_t checksum = 0;
  BB_FIELDS(copies, size, checksum)
};

struct BatchPutStartRequest {
  std::vector<PutStartRequest> requests;
  BB_FIELDS(requests)
};

struct BatchPutStartItem {
  int32_t status = 0;  // ErrorCode value
  std::vector<CopyPlacement> copies;
  BB_FIELDS(status, copies)
};

struct BatchPutStartResponse {
  std::vector<BatchPutStartItem> items;
  uint64_t view_version = 0;
  BB_FIELDS(items, view_version)
};

struct BatchGetWorkersItem {
  int32_t status = 0;
  GetWorkersResponse info;
  BB_FIELDS(status, info)
};

struct BatchGetWorkersResponse {
  std::vector<BatchGetWorkersItem> items;
  BB_FIELDS(items)
};

struct PingResponse {
  uint64_t view_version = 0;
  uint64_t server_time_ms = 0;
  bool is_leader = true;
  BB_FIELDS(view_version, server_time_ms, is_leader)
};

}  // namespace blackbird
