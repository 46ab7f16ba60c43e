// Bindings for the keystone / worker / client layers.
#include <map>
#include <thread>

#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "blackbird/client/client.h"
#include "blackbird/client/gpu_client.h"
#include "blackbird/client/shuffle.h"
#include "blackbird/gpu/gpu_kernels.h"
#include "blackbird/keystone/keystone_rpc.h"
#include "blackbird/keystone/keystone_service.h"
#include "blackbird/transport/rccl_engine.h"
#include "blackbird/worker/worker_service.h"

namespace py = pybind11;
using namespace blackbird;

namespace {

// Mirror of py_core.cpp's unwrap (kept local; the exception translator in
// py_core maps BlackbirdError for both TUs via a shared runtime_error).
template <typename T>
T unwrap(Result<T>&& r) {
  if (!r.ok())
    throw std::runtime_error(std::string(to_string(r.code())) +
                             (r.message().empty() ? "" : ": " + r.message()));
  return std::move(r.value());
}

inline void unwrap_void(Result<void>&& r) {
  if (!r.ok())
    throw std::runtime_error(std::string(to_string(r.code())) +
                             (r.message().empty() ? "" : ": " + r.message()));
}

}  // namespace

void bind_store(py::module_& m) {
  // ------------------------------------------------------------ configs
  py::class_<KeystoneConfig>(m, "KeystoneConfig")
      .def(py::init<>())
      .def_readwrite("cluster_id", &KeystoneConfig::cluster_id)
      .def_readwrite("listen_address", &KeystoneConfig::listen_address)
      .def_readwrite("coord_endpoint", &KeystoneConfig::coord_endpoint)
      .def_readwrite("metrics_address", &KeystoneConfig::metrics_address)
      .def_readwrite("object_ttl_default_ms", &KeystoneConfig::object_ttl_default_ms)
      .def_readwrite("gc_interval_ms", &KeystoneConfig::gc_interval_ms)
      .def_readwrite("health_interval_ms", &KeystoneConfig::health_interval_ms)
      .def_readwrite("worker_ttl_ms", &KeystoneConfig::worker_ttl_ms)
      .def_readwrite("eviction_high_watermark", &KeystoneConfig::eviction_high_watermark)
      .def_readwrite("eviction_ratio", &KeystoneConfig::eviction_ratio)
      .def_readwrite("enable_ha", &KeystoneConfig::enable_ha)
      .def_readwrite("enable_tiering", &KeystoneConfig::enable_tiering)
      .def_readwrite("tier_high_watermark", &KeystoneConfig::tier_high_watermark)
      .def_readwrite("tier_max_moves_per_cycle", &KeystoneConfig::tier_max_moves_per_cycle)
      .def_readwrite("promote_hot_threshold", &KeystoneConfig::promote_hot_threshold)
      .def_readwrite("compact_fragmentation_threshold",
                     &KeystoneConfig::compact_fragmentation_threshold)
      .def_readwrite("repair_max_per_cycle",
                     &KeystoneConfig::repair_max_per_cycle)
      .def_readwrite("scrub_interval_ms", &KeystoneConfig::scrub_interval_ms)
      .def_readwrite("scrub_batch", &KeystoneConfig::scrub_batch)
      .def_readwrite("persist_objects", &KeystoneConfig::persist_objects);

  py::class_<PoolConfig>(m, "PoolConfig")
      .def(py::init<>())
      .def_readwrite("pool_id", &PoolConfig::pool_id)
      .def_readwrite("storage_class", &PoolConfig::storage_class)
      .def_readwrite("size_bytes", &PoolConfig::size_bytes)
      .def_readwrite("mount_path", &PoolConfig::mount_path)
      .def_readwrite("gpu_device_id", &PoolConfig::gpu_device_id);

  py::class_<WorkerConfig>(m, "WorkerConfig")
      .def(py::init<>())
      .def_readwrite("worker_id", &WorkerConfig::worker_id)
      .def_readwrite("node_id", &WorkerConfig::node_id)
      .def_readwrite("cluster_id", &WorkerConfig::cluster_id)
      .def_readwrite("coord_endpoint", &WorkerConfig::coord_endpoint)
      .def_readwrite("data_listen_address", &WorkerConfig::data_listen_address)
      .def_readwrite("heartbeat_interval_ms", &WorkerConfig::heartbeat_interval_ms)
      .def_readwrite("heartbeat_ttl_ms", &WorkerConfig::heartbeat_ttl_ms)
      .def_readwrite("pools", &WorkerConfig::pools);

  py::class_<WorkerInfo>(m, "WorkerInfo")
      .def(py::init<>())
      .def_readwrite("worker_id", &WorkerInfo::worker_id)
      .def_readwrite("node_id", &WorkerInfo::node_id)
      .def_readwrite("data_endpoint", &WorkerInfo::data_endpoint)
      .def_readonly("last_heartbeat_ms", &WorkerInfo::last_heartbeat_ms);

  py::class_<ClusterStats>(m, "ClusterStats")
      .def_readonly("total_capacity", &ClusterStats::total_capacity)
      .def_readonly("total_used", &ClusterStats::total_used)
      .def_readonly("num_objects", &ClusterStats::num_objects)
      .def_readonly("num_workers", &ClusterStats::num_workers)
      .def_readonly("num_pools", &ClusterStats::num_pools)
      .def_readonly("view_version", &ClusterStats::view_version);

  py::class_<GetWorkersResponse>(m, "GetWorkersResponse")
      .def_readonly("copies", &GetWorkersResponse::copies)
      .def_readonly("size", &GetWorkersResponse::size)
      .def_readonly("checksum", &GetWorkersResponse::checksum);

  py::class_<PingResponse>(m, "PingResponse")
      .def_readonly("view_version", &PingResponse::view_version)
      .def_readonly("server_time_ms", &PingResponse::server_time_ms)
      .def_readonly("is_leader", &PingResponse::is_leader);

  // ------------------------------------------------------------ keystone
  py::class_<KeystoneService, std::shared_ptr<KeystoneService>>(m, "KeystoneService")
      .def(py::init([](const KeystoneConfig& cfg,
                       std::shared_ptr<coord::CoordService> coord) {
             return std::make_shared<KeystoneService>(cfg, std::move(coord));
           }),
           py::arg("config"), py::arg("coord") = nullptr)
      .def("initialize", [](KeystoneService& k) { unwrap_void(k.initialize()); })
      .def("start", [](KeystoneService& k) { unwrap_void(k.start()); },
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &KeystoneService::stop, py::call_guard<py::gil_scoped_release>())
      .def("object_exists", &KeystoneService::object_exists)
      .def("get_workers", [](KeystoneService& k, const std::string& key) {
        return unwrap(k.get_workers(key));
      })
      .def("put_start", [](KeystoneService& k, const std::string& key,
                           uint64_t size, const PlacementConfig& cfg) {
        auto r = unwrap(k.put_start(key, size, cfg));
        return r.copies;
      })
      .def("put_complete", [](KeystoneService& k, const std::string& key, uint64_t cs) {
        unwrap_void(k.put_complete(key, cs));
      }, py::arg("key"), py::arg("checksum") = 0)
      .def("put_cancel", [](KeystoneService& k, const std::string& key) {
        unwrap_void(k.put_cancel(key));
      })
      .def("remove_object", [](KeystoneService& k, const std::string& key) {
        unwrap_void(k.remove_object(key));
      })
      .def("remove_all_objects", &KeystoneService::remove_all_objects)
      .def("get_workers_info", &KeystoneService::get_workers_info)
      .def("get_memory_pools", &KeystoneService::get_memory_pools)
      .def("remove_worker", [](KeystoneService& k, const std::string& id) {
        unwrap_void(k.remove_worker(id));
      }, py::call_guard<py::gil_scoped_release>())
      .def("get_cluster_stats", &KeystoneService::get_cluster_stats)
      .def("get_view_version", &KeystoneService::get_view_version)
      .def("is_leader", &KeystoneService::is_leader)
      .def("register_pool", &KeystoneService::register_pool)
      .def("register_worker", &KeystoneService::register_worker)
      .def("batch_put_start_bench", [](KeystoneService& k,
                                        const std::vector<std::string>& keys,
                                        uint64_t size, const PlacementConfig& cfg) {
        std::vector<PutStartRequest> reqs;
        reqs.reserve(keys.size());
        for (auto& key : keys) reqs.push_back({key, size, cfg});
        py::gil_scoped_release rel;
        auto resp = k.batch_put_start(reqs);
        int ok = 0;
        for (auto& it : resp.items)
          if (it.status == 0) ++ok;
        return ok;
      })
      .def("batch_remove_bench", [](KeystoneService& k,
                                    const std::vector<std::string>& keys) {
        py::gil_scoped_release rel;
        auto st = k.batch_remove(keys);
        return static_cast<int>(st.size());
      })
      .def("run_gc_once", &KeystoneService::run_gc_once)
      .def("run_eviction_once", &KeystoneService::run_eviction_once)
      .def("run_tiering_once", &KeystoneService::run_tiering_once,
           py::call_guard<py::gil_scoped_release>())
      .def("migrate_object", [](KeystoneService& k, const std::string& key,
                                StorageClass target) {
        unwrap_void(k.migrate_object(key, target));
      }, py::call_guard<py::gil_scoped_release>())
      .def("run_repair_once", &KeystoneService::run_repair_once,
           py::call_guard<py::gil_scoped_release>())
      .def("compact_pool", [](KeystoneService& k, const std::string& pool,
                              uint32_t max_moves) {
        return unwrap(k.compact_pool(pool, max_moves));
      }, py::arg("pool_id"), py::arg("max_moves") = 64,
         py::call_guard<py::gil_scoped_release>())
      .def("run_compaction_once", &KeystoneService::run_compaction_once,
           py::call_guard<py::gil_scoped_release>())
      .def("run_scrub_once", &KeystoneService::run_scrub_once,
           py::arg("max_objects") = 0,
           py::call_guard<py::gil_scoped_release>())
      .def("counters", [](KeystoneService& k) {
        auto c = k.counters();
        py::dict d;
        d["migrations"] = c.migrations;
        d["repairs"] = c.repairs;
        d["scrub_quarantined"] = c.scrub_quarantined;
        d["evictions"] = c.evictions;
        d["gc_reclaimed"] = c.gc_reclaimed;
        return d;
      })
      .def("repair_object", [](KeystoneService& k, const std::string& key) {
        unwrap_void(k.repair_object(key));
      }, py::call_guard<py::gil_scoped_release>())
      // sessionful upsert protocol (CPU-testable surface)
      .def("create_put_session", [](KeystoneService& k,
                                    const std::vector<std::string>& keys,
                                    uint64_t size, const PlacementConfig& cfg) {
        std::vector<PutStartRequest> reqs;
        reqs.reserve(keys.size());
        for (auto& key : keys) reqs.push_back({key, size, cfg});
        return k.create_put_session(reqs);
      })
      .def("upsert_start_token", [](KeystoneService& k, uint64_t token) {
        unwrap_void(k.upsert_start_token(token));
      })
      .def("commit_token", [](KeystoneService& k, uint64_t token,
                              const std::vector<uint64_t>& digests) {
        unwrap_void(k.commit_token(token, digests));
      })
      .def("token_commits", &KeystoneService::token_commits)
      .def("coord", &KeystoneService::coord);

  py::class_<KeystoneServer, std::shared_ptr<KeystoneServer>>(m, "KeystoneServer")
      .def(py::init<std::shared_ptr<KeystoneService>>())
      .def("start", [](KeystoneServer& s) { unwrap_void(s.start()); })
      .def("stop", &KeystoneServer::stop, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &KeystoneServer::port)
      .def_property_readonly("endpoint", &KeystoneServer::endpoint)
      .def_property_readonly("metrics_port", &KeystoneServer::metrics_port)
      .def("service", &KeystoneServer::service);

  m.def("create_and_start_keystone",
        [](const KeystoneConfig& cfg, std::shared_ptr<coord::CoordService> coord) {
          return unwrap(create_and_start_keystone(cfg, std::move(coord)));
        },
        py::arg("config"), py::arg("coord") = nullptr,
        py::call_guard<py::gil_scoped_release>());

  // ------------------------------------------------------------- worker
  py::class_<StorageStats>(m, "StorageStats")
      .def_readonly("capacity", &StorageStats::capacity)
      .def_readonly("used", &StorageStats::used)
      .def_readonly("reserved", &StorageStats::reserved)
      .def_readonly("num_shards", &StorageStats::num_shards)
      .def_readonly("num_reservations", &StorageStats::num_reservations);

  py::class_<ReservationToken>(m, "ReservationToken")
      .def_readonly("token_id", &ReservationToken::token_id)
      .def_readonly("offset", &ReservationToken::offset)
      .def_readonly("size", &ReservationToken::size)
      .def_readonly("expires_ms", &ReservationToken::expires_ms);

  py::class_<StorageBackend>(m, "StorageBackend")
      .def("storage_class", &StorageBackend::storage_class)
      .def("capacity", &StorageBackend::capacity)
      .def("access_info", &StorageBackend::access_info)
      .def("reserve", [](StorageBackend& b, uint64_t size) {
        return unwrap(b.reserve(size));
      })
      .def("reserve_at", [](StorageBackend& b, uint64_t off, uint64_t size) {
        return unwrap(b.reserve_at(off, size));
      })
      .def("commit", [](StorageBackend& b, uint64_t t) { unwrap_void(b.commit(t)); })
      .def("abort", [](StorageBackend& b, uint64_t t) { unwrap_void(b.abort(t)); })
      .def("free", [](StorageBackend& b, uint64_t off, uint64_t size) {
        unwrap_void(b.free(off, size));
      })
      .def("write", [](StorageBackend& b, uint64_t off, py::buffer buf) {
        py::buffer_info info = buf.request();
        unwrap_void(b.write(off, info.ptr,
                            static_cast<uint64_t>(info.size * info.itemsize)));
      })
      .def("read", [](StorageBackend& b, uint64_t off, uint64_t len) {
        std::string out;
        out.resize(len);
        unwrap_void(b.read(off, out.data(), len));
        return py::bytes(out);
      })
      .def("checksum", [](StorageBackend& b, uint64_t off, uint64_t len) {
        return unwrap(b.checksum(off, len));
      })
      .def("stats", &StorageBackend::stats);

  m.def("make_backend", [](const PoolConfig& cfg, const std::string& worker_id) {
    auto b = unwrap(create_storage_backend(cfg, worker_id));
    unwrap_void(b->initialize());
    return b;
  });

  py::class_<WorkerService>(m, "WorkerService")
      .def(py::init([](const WorkerConfig& cfg,
                       std::shared_ptr<coord::CoordService> coord) {
             return std::make_unique<WorkerService>(cfg, std::move(coord));
           }),
           py::arg("config"), py::arg("coord") = nullptr)
      .def("initialize", [](WorkerService& w) { unwrap_void(w.initialize()); },
           py::call_guard<py::gil_scoped_release>())
      .def("start", [](WorkerService& w) { unwrap_void(w.start()); },
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &WorkerService::stop, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("data_endpoint", &WorkerService::data_endpoint)
      .def("pool_descriptors", &WorkerService::pool_descriptors)
      .def("stats_json", &WorkerService::stats_json)
      .def("backend", &WorkerService::backend, py::return_value_policy::reference_internal);

  // ------------------------------------------------------------- client
  py::class_<ClientOptions>(m, "ClientOptions")
      .def(py::init<>())
      .def_readwrite("keystone_endpoint", &ClientOptions::keystone_endpoint)
      .def_readwrite("coord_endpoint", &ClientOptions::coord_endpoint)
      .def_readwrite("io_threads", &ClientOptions::io_threads)
      .def_readwrite("verify_checksum_on_get", &ClientOptions::verify_checksum_on_get)
      .def_readwrite("rpc_timeout_ms", &ClientOptions::rpc_timeout_ms)
      .def_readwrite("force_tcp", &ClientOptions::force_tcp);

  py::class_<Client::HostPutSession>(m, "HostPutSession")
      .def(py::init<>())
      .def_property_readonly(
          "active", [](const Client::HostPutSession& s) { return s.token != 0; });

  py::class_<Client>(m, "Client")
      .def(py::init<ClientOptions>(), py::arg("options") = ClientOptions{})
      .def("connect", [](Client& c) { unwrap_void(c.connect()); },
           py::call_guard<py::gil_scoped_release>())
      .def("close", &Client::close, py::call_guard<py::gil_scoped_release>())
      .def("put", [](Client& c, const std::string& key, py::buffer buf,
                     const PlacementConfig& cfg) {
        py::buffer_info info = buf.request();
        py::gil_scoped_release rel;
        unwrap_void(c.put(key, info.ptr,
                          static_cast<uint64_t>(info.size * info.itemsize), cfg));
      }, py::arg("key"), py::arg("data"), py::arg("config") = PlacementConfig{})
      .def("get", [](Client& c, const std::string& key) {
        std::string out;
        {
          py::gil_scoped_release rel;
          out = unwrap(c.get(key));
        }
        return py::bytes(out);
      })
      .def("exists", [](Client& c, const std::string& key) {
        return unwrap(c.exists(key));
      }, py::call_guard<py::gil_scoped_release>())
      .def("remove", [](Client& c, const std::string& key) {
        unwrap_void(c.remove(key));
      }, py::call_guard<py::gil_scoped_release>())
      .def("remove_all", [](Client& c) { return unwrap(c.remove_all()); },
           py::call_guard<py::gil_scoped_release>())
      .def("batch_remove", [](Client& c, const std::vector<std::string>& keys) {
        return unwrap(c.batch_remove(keys));
      }, py::call_guard<py::gil_scoped_release>())
      .def("batch_put", [](Client& c, const std::vector<std::pair<std::string, py::buffer>>& items,
                           const PlacementConfig& cfg) {
        std::vector<Client::PutItem> its;
        std::vector<py::buffer_info> infos;
        its.reserve(items.size());
        infos.reserve(items.size());
        for (auto& [k, b] : items) {
          infos.push_back(const_cast<py::buffer&>(b).request());
          its.push_back({k, infos.back().ptr,
                         static_cast<uint64_t>(infos.back().size * infos.back().itemsize)});
        }
        py::gil_scoped_release rel;
        return unwrap(c.batch_put(its, cfg));
      }, py::arg("items"), py::arg("config") = PlacementConfig{})
      .def("batch_put_session",
           [](Client& c,
              const std::vector<std::pair<std::string, py::buffer>>& items,
              const PlacementConfig& cfg, Client::HostPutSession& sess) {
             std::vector<Client::PutItem> its;
             std::vector<py::buffer_info> infos;
             its.reserve(items.size());
             infos.reserve(items.size());
             for (auto& [k, b] : items) {
               infos.push_back(const_cast<py::buffer&>(b).request());
               its.push_back({k, infos.back().ptr,
                              static_cast<uint64_t>(infos.back().size *
                                                    infos.back().itemsize)});
             }
             py::gil_scoped_release rel;
             return unwrap(c.batch_put(its, cfg, &sess));
           },
           py::arg("items"), py::arg("config"), py::arg("session"))
      .def_property_readonly("host_session_steps",
                             &Client::host_session_steps)
      .def("batch_get", [](Client& c, const std::vector<std::string>& keys) {
        std::vector<std::pair<int32_t, std::string>> res;
        {
          py::gil_scoped_release rel;
          res = unwrap(c.batch_get(keys));
        }
        py::list out;
        for (auto& [status, data] : res)
          out.append(py::make_tuple(status, py::bytes(data)));
        return out;
      })
      .def("cluster_stats", [](Client& c) { return unwrap(c.cluster_stats()); },
           py::call_guard<py::gil_scoped_release>())
      .def("memory_pools", [](Client& c) { return unwrap(c.memory_pools()); },
           py::call_guard<py::gil_scoped_release>())
      .def("list_objects", [](Client& c, const std::string& prefix,
                              uint32_t limit) {
        std::vector<ObjectSummary> rows;
        {
          py::gil_scoped_release rel;
          rows = unwrap(c.list_objects(prefix, limit));
        }
        py::list out;
        for (auto& o : rows)
          out.append(py::make_tuple(o.key, o.size, o.ncopies,
                                    o.storage_class));
        return out;
      }, py::arg("prefix") = "", py::arg("limit") = 1000)
      .def("workers_info", [](Client& c) { return unwrap(c.workers_info()); },
           py::call_guard<py::gil_scoped_release>())
      .def("ping", [](Client& c) { return unwrap(c.ping()); },
           py::call_guard<py::gil_scoped_release>());

  // ---------------------------------------------- prepared batches
  // Conversion of thousands of Python tuples per call dominates small-object
  // batch latency; prepared batches convert once and are reused every step.
  struct DevPutBatch {
    std::vector<GpuClient::DevPutItem> items;
    std::vector<std::string> keys;
    GpuClient::BatchPutSession session;  // steady-state token fast path
  };
  struct DevGetBatch {
    std::vector<GpuClient::DevGetItem> items;
    GpuClient::BatchGetSession session;
  };
  py::class_<DevPutBatch>(m, "DevPutBatch")
      .def_property_readonly("size", [](const DevPutBatch& b) { return b.items.size(); });
  py::class_<DevGetBatch>(m, "DevGetBatch")
      .def_property_readonly("size", [](const DevGetBatch& b) { return b.items.size(); });
  m.def("make_put_batch",
        [](const std::vector<std::tuple<std::string, uint64_t, uint64_t>>& items) {
          DevPutBatch b;
          for (auto& [k, p, sz] : items) {
            b.items.push_back({k, reinterpret_cast<const void*>(p), sz});
            b.keys.push_back(k);
          }
          return b;
        });
  m.def("make_get_batch",
        [](const std::vector<std::tuple<std::string, uint64_t, uint64_t>>& items) {
          DevGetBatch b;
          for (auto& [k, p, sz] : items)
            b.items.push_back({k, reinterpret_cast<void*>(p), sz});
          return b;
        });

  // -------------------------------------------------------- gpu client
  py::class_<GpuClient>(m, "GpuClient")
      .def(py::init<Client&, int>(), py::arg("client"), py::arg("device") = 0,
           py::keep_alive<1, 2>())
      .def("init", [](GpuClient& g) { unwrap_void(g.init()); },
           py::call_guard<py::gil_scoped_release>())
      .def("set_fused_copy", &GpuClient::set_fused_copy)
      .def("put_device", [](GpuClient& g, const std::string& key, uint64_t ptr,
                            uint64_t size, const PlacementConfig& cfg) {
        unwrap_void(g.put_device(key, reinterpret_cast<const void*>(ptr), size, cfg));
      }, py::arg("key"), py::arg("dev_ptr"), py::arg("size"),
         py::arg("config") = PlacementConfig{},
         py::call_guard<py::gil_scoped_release>())
      .def("get_device", [](GpuClient& g, const std::string& key, uint64_t ptr,
                            uint64_t cap, bool verify) {
        return unwrap(g.get_device(key, reinterpret_cast<void*>(ptr), cap, verify));
      }, py::arg("key"), py::arg("dev_ptr"), py::arg("capacity"),
         py::arg("verify") = false,
         py::call_guard<py::gil_scoped_release>())
      .def("batch_put_device",
           [](GpuClient& g,
              const std::vector<std::tuple<std::string, uint64_t, uint64_t>>& items,
              const PlacementConfig& cfg) {
             std::vector<GpuClient::DevPutItem> its;
             for (auto& [k, p, s] : items)
               its.push_back({k, reinterpret_cast<const void*>(p), s});
             py::gil_scoped_release rel;
             return unwrap(g.batch_put_device(its, cfg));
           }, py::arg("items"), py::arg("config") = PlacementConfig{})
      .def("batch_get_device",
           [](GpuClient& g,
              const std::vector<std::tuple<std::string, uint64_t, uint64_t>>& items,
              bool verify) {
             std::vector<GpuClient::DevGetItem> its;
             for (auto& [k, p, s] : items)
               its.push_back({k, reinterpret_cast<void*>(p), s});
             py::gil_scoped_release rel;
             return unwrap(g.batch_get_device(its, verify));
           }, py::arg("items"), py::arg("verify") = false)
      .def("batch_put_prepared",
           [](GpuClient& g, DevPutBatch& b, const PlacementConfig& cfg) {
             py::gil_scoped_release rel;
             auto st = unwrap(g.batch_put_device(b.items, cfg, &b.session));
             for (auto v : st)
               if (v != 0) return false;
             return true;
           }, py::arg("batch"), py::arg("config") = PlacementConfig{})
      .def("batch_get_prepared",
           [](GpuClient& g, DevGetBatch& b, bool verify) {
             py::gil_scoped_release rel;
             auto st = unwrap(g.batch_get_device(b.items, verify, &b.session));
             for (auto v : st)
               if (v != 0) return false;
             return true;
           }, py::arg("batch"), py::arg("verify") = false)
      .def_property_readonly("session_put_steps", &GpuClient::session_put_steps)
      .def_property_readonly("session_get_steps", &GpuClient::session_get_steps)
      .def_property_readonly("session_graph_steps",
                             &GpuClient::session_graph_steps)
      .def("batch_put_async",
           [](GpuClient& g,
              const std::vector<std::tuple<std::string, uint64_t, uint64_t>>& items,
              const PlacementConfig& cfg) {
             std::vector<GpuClient::DevPutItem> its;
             for (auto& [k, p, s] : items)
               its.push_back({k, reinterpret_cast<const void*>(p), s});
             py::gil_scoped_release rel;
             return unwrap(g.batch_put_async(std::move(its), cfg));
           }, py::arg("items"), py::arg("config") = PlacementConfig{})
      .def("batch_get_async",
           [](GpuClient& g,
              const std::vector<std::tuple<std::string, uint64_t, uint64_t>>& items,
              bool verify) {
             std::vector<GpuClient::DevGetItem> its;
             for (auto& [k, p, s] : items)
               its.push_back({k, reinterpret_cast<void*>(p), s});
             py::gil_scoped_release rel;
             return unwrap(g.batch_get_async(std::move(its), verify));
           }, py::arg("items"), py::arg("verify") = false)
      .def("async_wait", [](GpuClient& g, uint64_t token) {
        return unwrap(g.async_wait(token));
      }, py::arg("token"), py::call_guard<py::gil_scoped_release>())
      .def("set_placement_cache", &GpuClient::set_placement_cache)
      .def("invalidate", &GpuClient::invalidate,
           py::call_guard<py::gil_scoped_release>())
      .def("clear_placement_cache", &GpuClient::clear_placement_cache);

  // ------------------------------------------------------- rccl engine
  py::class_<RcclEngine>(m, "RcclEngine")
      .def(py::init<>())
      .def("init", [](RcclEngine& e, std::shared_ptr<coord::CoordService> coord,
                      const std::string& cluster, const std::string& tag,
                      int rank, int nranks, int device, int timeout_ms) {
        unwrap_void(e.init(std::move(coord), cluster, tag, rank, nranks, device,
                           timeout_ms));
      }, py::arg("coord"), py::arg("cluster_id"), py::arg("tag"),
         py::arg("rank"), py::arg("nranks"), py::arg("device"),
         py::arg("timeout_ms") = 60000,
         py::call_guard<py::gil_scoped_release>())
      .def("destroy", &RcclEngine::destroy,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("rank", &RcclEngine::rank)
      .def_property_readonly("nranks", &RcclEngine::nranks)
      .def("alltoallv", [](RcclEngine& e, const std::vector<uint64_t>& send_ptrs,
                           const std::vector<uint64_t>& send_bytes,
                           const std::vector<uint64_t>& recv_ptrs,
                           const std::vector<uint64_t>& recv_bytes) {
        std::vector<const void*> sp;
        std::vector<void*> rp;
        for (auto p : send_ptrs) sp.push_back(reinterpret_cast<const void*>(p));
        for (auto p : recv_ptrs) rp.push_back(reinterpret_cast<void*>(p));
        py::gil_scoped_release rel;
        unwrap_void(e.alltoallv(sp, send_bytes, rp, recv_bytes, nullptr));
        unwrap_void(gpu::sync());
      })
      .def("send", [](RcclEngine& e, uint64_t ptr, uint64_t n, int peer) {
        py::gil_scoped_release rel;
        unwrap_void(e.send(reinterpret_cast<const void*>(ptr), n, peer, nullptr));
        unwrap_void(gpu::sync());
      })
      .def("recv", [](RcclEngine& e, uint64_t ptr, uint64_t n, int peer) {
        py::gil_scoped_release rel;
        unwrap_void(e.recv(reinterpret_cast<void*>(ptr), n, peer, nullptr));
        unwrap_void(gpu::sync());
      });

  // ------------------------------------------- collective batch shuffle
  // wants: per peer (keys, sizes, recv_base device ptr)
  m.def("gpu_batch_shuffle",
        [](GpuClient& g, RcclEngine& e,
           const std::vector<std::tuple<std::vector<std::string>,
                                        std::vector<uint64_t>, uint64_t>>& wants) {
          std::vector<ShuffleWant> w;
          w.reserve(wants.size());
          for (auto& [keys, sizes, base] : wants) {
            ShuffleWant sw;
            sw.keys = keys;
            sw.sizes = sizes;
            sw.recv_base = reinterpret_cast<void*>(base);
            w.push_back(std::move(sw));
          }
          py::gil_scoped_release rel;
          unwrap_void(g.batch_shuffle_rccl(e, w));
        });

  // Host-loopback shuffle (tests): N in-process "ranks" exchange through
  // shared memory — exercises the full batch_shuffle algorithm (want-list
  // encode/exchange, local resolution, gather layout, consistent-abort
  // handshake, data all-to-all) with no GPU.
  // objects[r]: key → bytes that rank r OWNS.
  // want_keys[r][p]: keys rank r wants from rank p.
  // Returns recv[r][p]: the concatenated payload rank r received from p.
  m.def("loopback_shuffle_for_test",
        [](const std::vector<std::map<std::string, py::bytes>>& objects,
           const std::vector<std::vector<std::vector<std::string>>>& want_keys) {
          const int n = static_cast<int>(objects.size());
          // materialize owned objects as host buffers
          std::vector<std::map<std::string, std::string>> own(n);
          for (int r = 0; r < n; ++r)
            for (auto& [k, v] : objects[r]) own[r][k] = std::string(v);
          LoopbackGroup grp(n);
          std::vector<std::vector<std::string>> recv(n);      // per rank, per peer
          std::vector<std::vector<ShuffleWant>> wants(n);
          for (int r = 0; r < n; ++r) {
            recv[r].resize(n);
            wants[r].resize(n);
            for (int p = 0; p < n; ++p) {
              auto& w = wants[r][p];
              w.keys = want_keys[r][p];
              uint64_t total = 0;
              for (auto& k : w.keys) {
                // sizes must be known to the requester (fixed-size pattern):
                // look them up in the global object table
                uint64_t sz = 0;
                for (int q = 0; q < n; ++q) {
                  auto it = own[q].find(k);
                  if (it != own[q].end()) sz = it->second.size();
                }
                w.sizes.push_back(sz);
                total += sz;
              }
              recv[r][p].resize(total);
              w.recv_base = recv[r][p].data();
            }
          }
          std::vector<std::string> errs(n);
          {
            py::gil_scoped_release rel;
            std::vector<std::thread> ts;
            for (int r = 0; r < n; ++r)
              ts.emplace_back([&, r] {
                auto ex = grp.exchanger(r);
                HostCopier cp;
                ShuffleResolver res = [&, r](const ObjectKey& k,
                                             uint64_t sz) -> const void* {
                  auto it = own[r].find(k);
                  if (it == own[r].end() || it->second.size() != sz)
                    return nullptr;
                  return it->second.data();
                };
                auto rr = batch_shuffle(*ex, cp, res, wants[r]);
                if (!rr.ok()) errs[r] = rr.message();
              });
            for (auto& t : ts) t.join();
          }
          for (int r = 0; r < n; ++r)
            if (!errs[r].empty())
              throw std::runtime_error("rank " + std::to_string(r) + ": " +
                                       errs[r]);
          std::vector<std::vector<py::bytes>> out(n);
          for (int r = 0; r < n; ++r)
            for (int p = 0; p < n; ++p)
              out[r].emplace_back(recv[r][p]);
          return out;
        });

  m.def("client_batch_remove_prepared", [](Client& c, const DevPutBatch& b) {
    py::gil_scoped_release rel;
    auto st = unwrap(c.batch_remove(b.keys));
    for (auto v : st)
      if (v != 0) return false;
    return true;
  });
}
