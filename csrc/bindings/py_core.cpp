// pybind11 bindings: the Python surface of the MI355X-native object store.
// Python is the test/benchmark harness; all logic lives in the C++/HIP core.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "blackbird/allocation/pool_allocator.h"
#include "blackbird/allocation/range_allocator.h"
#include "blackbird/common/result.h"
#include "blackbird/common/types.h"
#include "blackbird/coord/coord.h"
#include "blackbird/gpu/gpu_kernels.h"

namespace py = pybind11;
using namespace blackbird;

namespace {

class BlackbirdError : public std::runtime_error {
 public:
  BlackbirdError(ErrorCode code, const std::string& msg)
      : std::runtime_error(std::string(to_string(code)) +
                           (msg.empty() ? "" : (": " + msg))),
        code_(code) {}
  ErrorCode code_;
};

template <typename T>
T unwrap(Result<T>&& r) {
  if (!r.ok()) throw BlackbirdError(r.code(), r.message());
  return std::move(r.value());
}

inline void unwrap_void(Result<void>&& r) {
  if (!r.ok()) throw BlackbirdError(r.code(), r.message());
}

}  // namespace

void bind_store(py::module_& m);  // keystone/worker/client (py_store.cpp)

PYBIND11_MODULE(_core, m) {
  m.doc() = "blackbird_amd core — MI355X-native tiered distributed object store";

  static py::exception<BlackbirdError> exc(m, "BlackbirdError");
  py::register_exception_translator([](std::exception_ptr p) {
    try {
      if (p) std::rethrow_exception(p);
    } catch (const BlackbirdError& e) {
      py::set_error(exc, e.what());
    }
  });

  // ------------------------------------------------------------ enums
  py::enum_<StorageClass>(m, "StorageClass")
      .value("RAM_CPU", StorageClass::RAM_CPU)
      .value("RAM_GPU", StorageClass::RAM_GPU)
      .value("PINNED_CPU", StorageClass::PINNED_CPU)
      .value("NVME", StorageClass::NVME)
      .value("SSD", StorageClass::SSD)
      .value("HDD", StorageClass::HDD)
      .value("CXL_MEM", StorageClass::CXL_MEM);

  py::enum_<AccessKind>(m, "AccessKind")
      .value("TCP", AccessKind::TCP)
      .value("SHM", AccessKind::SHM)
      .value("HIP_IPC", AccessKind::HIP_IPC);

  // ------------------------------------------------------------ structs
  py::class_<AccessInfo>(m, "AccessInfo")
      .def(py::init<>())
      .def_readwrite("kind", &AccessInfo::kind)
      .def_readwrite("endpoint", &AccessInfo::endpoint)
      .def_readwrite("shm_name", &AccessInfo::shm_name)
      .def_readwrite("device_id", &AccessInfo::device_id)
      .def_readwrite("ipc_handle_hex", &AccessInfo::ipc_handle_hex)
      .def_readwrite("base_addr", &AccessInfo::base_addr);

  py::class_<MemoryPool>(m, "MemoryPool")
      .def(py::init<>())
      .def_readwrite("pool_id", &MemoryPool::pool_id)
      .def_readwrite("worker_id", &MemoryPool::worker_id)
      .def_readwrite("node_id", &MemoryPool::node_id)
      .def_readwrite("storage_class", &MemoryPool::storage_class)
      .def_readwrite("size", &MemoryPool::size)
      .def_readwrite("used", &MemoryPool::used)
      .def_readwrite("access", &MemoryPool::access)
      .def("to_json", [](const MemoryPool& p) { return p.to_json().dump(); })
      .def_static("from_json", [](const std::string& s) {
        return MemoryPool::from_json(json::parse_or_null(s));
      });

  py::class_<ShardPlacement>(m, "ShardPlacement")
      .def(py::init<>())
      .def_readwrite("pool_id", &ShardPlacement::pool_id)
      .def_readwrite("worker_id", &ShardPlacement::worker_id)
      .def_readwrite("storage_class", &ShardPlacement::storage_class)
      .def_readwrite("offset", &ShardPlacement::offset)
      .def_readwrite("length", &ShardPlacement::length)
      .def_readwrite("access", &ShardPlacement::access)
      .def_readwrite("digest", &ShardPlacement::digest);

  py::class_<CopyPlacement>(m, "CopyPlacement")
      .def(py::init<>())
      .def_readwrite("copy_index", &CopyPlacement::copy_index)
      .def_readwrite("shards", &CopyPlacement::shards);

  py::class_<PlacementConfig>(m, "PlacementConfig")
      .def(py::init<>())
      .def_readwrite("replication", &PlacementConfig::replication)
      .def_readwrite("max_workers_per_copy", &PlacementConfig::max_workers_per_copy)
      .def_readwrite("min_shard_size", &PlacementConfig::min_shard_size)
      .def_readwrite("preferred_class", &PlacementConfig::preferred_class)
      .def_readwrite("required_class", &PlacementConfig::required_class)
      .def_readwrite("ttl_ms", &PlacementConfig::ttl_ms)
      .def_readwrite("checksum", &PlacementConfig::checksum)
      .def_readwrite("preferred_worker", &PlacementConfig::preferred_worker)
      .def_readwrite("replace", &PlacementConfig::replace);

  py::class_<PoolAllocatorStats>(m, "PoolAllocatorStats")
      .def_readonly("capacity", &PoolAllocatorStats::capacity)
      .def_readonly("used", &PoolAllocatorStats::used)
      .def_readonly("free_ranges", &PoolAllocatorStats::free_ranges)
      .def_readonly("largest_free", &PoolAllocatorStats::largest_free)
      .def_readonly("fragmentation", &PoolAllocatorStats::fragmentation);

  py::class_<AllocatorStats>(m, "AllocatorStats")
      .def_readonly("total_capacity", &AllocatorStats::total_capacity)
      .def_readonly("total_used", &AllocatorStats::total_used)
      .def_readonly("num_pools", &AllocatorStats::num_pools)
      .def_readonly("num_objects", &AllocatorStats::num_objects)
      .def_readonly("fragmentation", &AllocatorStats::fragmentation);

  // --------------------------------------------------------- allocators
  py::class_<PoolAllocator>(m, "PoolAllocator")
      .def(py::init([](uint64_t cap, const std::string& policy, uint64_t align) {
             return std::make_unique<PoolAllocator>(
                 cap,
                 policy == "first_fit" ? PoolAllocator::Policy::FIRST_FIT
                                       : PoolAllocator::Policy::BEST_FIT,
                 align);
           }),
           py::arg("capacity"), py::arg("policy") = "best_fit",
           py::arg("alignment") = 256)
      .def("allocate", [](PoolAllocator& a, uint64_t size) {
        return unwrap(a.allocate(size));
      })
      .def("free", [](PoolAllocator& a, uint64_t off, uint64_t size) {
        unwrap_void(a.free(off, size));
      })
      .def("reserve_exact", [](PoolAllocator& a, uint64_t off, uint64_t size) {
        unwrap_void(a.reserve_exact(off, size));
      })
      .def("used", &PoolAllocator::used)
      .def("available", &PoolAllocator::available)
      .def_property_readonly("capacity", &PoolAllocator::capacity)
      .def("stats", &PoolAllocator::stats);

  py::class_<RangeAllocator>(m, "RangeAllocator")
      .def(py::init<>())
      .def("upsert_pool", &RangeAllocator::upsert_pool)
      .def("remove_pool", &RangeAllocator::remove_pool)
      .def("pools", &RangeAllocator::pools)
      .def("allocate", [](RangeAllocator& a, const std::string& key, uint64_t size,
                          const PlacementConfig& cfg) {
        return unwrap(a.allocate(key, size, cfg));
      })
      .def("free", [](RangeAllocator& a, const std::string& key) {
        unwrap_void(a.free(key));
      })
      .def("can_allocate", &RangeAllocator::can_allocate)
      .def("allocate_batch", [](RangeAllocator& a,
                                const std::vector<std::string>& keys,
                                const std::vector<uint64_t>& sizes,
                                const PlacementConfig& cfg) {
        auto out = a.allocate_batch(keys, sizes, cfg);
        py::list res;
        for (auto& [st, copies] : out)
          res.append(py::make_tuple(st, copies));
        return res;
      })
      .def("allocate_batch_bench", [](RangeAllocator& a,
                                      const std::vector<std::string>& keys,
                                      uint64_t size, const PlacementConfig& cfg) {
        std::vector<uint64_t> sizes(keys.size(), size);
        py::gil_scoped_release rel;
        auto out = a.allocate_batch(keys, sizes, cfg);
        int ok = 0;
        for (auto& [st, c] : out)
          if (st == 0) ++ok;
        return ok;
      })
      .def("free_keys_bench", [](RangeAllocator& a,
                                 const std::vector<std::string>& keys) {
        py::gil_scoped_release rel;
        for (auto& k : keys) a.free(k);
      })
      .def("stats", &RangeAllocator::stats);

  // --------------------------------------------------------- coordination
  py::enum_<coord::EventType>(m, "EventType")
      .value("PUT", coord::EventType::PUT)
      .value("DELETE", coord::EventType::DELETE)
      .value("EXPIRE", coord::EventType::EXPIRE);

  py::class_<coord::WatchEvent>(m, "WatchEvent")
      .def_readonly("type", &coord::WatchEvent::type)
      .def_readonly("key", &coord::WatchEvent::key)
      .def_readonly("value", &coord::WatchEvent::value);

  py::class_<coord::CoordStore, std::shared_ptr<coord::CoordStore>>(m, "CoordStore")
      .def(py::init<>())
      .def("sweep_now", &coord::CoordStore::sweep_now)
      .def("size", &coord::CoordStore::size)
      .def("put", [](coord::CoordStore& s, const std::string& k,
                     const std::string& v, uint64_t ttl_ms) {
        unwrap_void(s.put(k, v, ttl_ms));
      }, py::arg("key"), py::arg("value"), py::arg("ttl_ms") = 0)
      .def("get", [](coord::CoordStore& s, const std::string& k) {
        return unwrap(s.get(k));
      })
      .def("save", [](coord::CoordStore& s, const std::string& path) {
        unwrap_void(s.save(path));
      })
      .def("load", [](coord::CoordStore& s, const std::string& path) {
        unwrap_void(s.load(path));
      })
      .def("get_prefix",
           [](coord::CoordStore& s, const std::string& p) {
             auto r = s.get_prefix(p);
             if (!r.ok()) throw std::runtime_error(r.error().message);
             std::vector<std::pair<std::string, py::bytes>> out;
             for (auto& kv : r.value())
               out.emplace_back(kv.key, py::bytes(kv.value));
             return out;
           })
      .def("dirty", &coord::CoordStore::dirty)
      .def("epoch", &coord::CoordStore::epoch)
      .def("bump_epoch", &coord::CoordStore::bump_epoch);

  py::class_<coord::CoordService, std::shared_ptr<coord::CoordService>>(m, "CoordService")
      .def("put", [](coord::CoordService& c, const std::string& k,
                     const std::string& v, uint64_t ttl) {
        unwrap_void(c.put(k, v, ttl));
      }, py::arg("key"), py::arg("value"), py::arg("ttl_ms") = 0,
         py::call_guard<py::gil_scoped_release>())
      .def("get", [](coord::CoordService& c, const std::string& k) {
        return unwrap(c.get(k));
      }, py::call_guard<py::gil_scoped_release>())
      .def("delete_", [](coord::CoordService& c, const std::string& k) {
        unwrap_void(c.del(k));
      }, py::call_guard<py::gil_scoped_release>())
      .def("get_prefix", [](coord::CoordService& c, const std::string& p) {
        auto kvs = unwrap(c.get_prefix(p));
        std::vector<std::pair<std::string, std::string>> out;
        for (auto& kv : kvs) out.emplace_back(kv.key, kv.value);
        return out;
      }, py::call_guard<py::gil_scoped_release>())
      .def("cas", [](coord::CoordService& c, const std::string& k,
                     const std::string& expected, bool expect_absent,
                     const std::string& v, uint64_t ttl) {
        return unwrap(c.cas(k, expected, expect_absent, v, ttl));
      }, py::arg("key"), py::arg("expected"), py::arg("expect_absent"),
         py::arg("value"), py::arg("ttl_ms") = 0,
         py::call_guard<py::gil_scoped_release>())
      .def("keep_alive", [](coord::CoordService& c, const std::string& k, uint64_t ttl) {
        unwrap_void(c.keep_alive(k, ttl));
      }, py::call_guard<py::gil_scoped_release>())
      .def("watch_prefix", [](coord::CoordService& c, const std::string& p,
                              py::function cb) {
        // callback invoked from C++ threads → GIL discipline: hold the
        // py::function behind a shared_ptr (C++-side copies must not touch
        // refcounts) and delete it under the GIL.
        std::shared_ptr<py::function> fptr(
            new py::function(std::move(cb)), [](py::function* f) {
              py::gil_scoped_acquire g;
              delete f;
            });
        auto wrapped = [fptr](const coord::WatchEvent& ev) {
          py::gil_scoped_acquire g;
          try {
            (*fptr)(ev);
          } catch (py::error_already_set& e) {
            e.discard_as_unraisable("watch callback");
          }
        };
        return unwrap(c.watch_prefix(p, wrapped));
      })  // keeps GIL: only registers; RPC inside is fast
      .def("unwatch", [](coord::CoordService& c, uint64_t id) {
        unwrap_void(c.unwatch(id));
      }, py::call_guard<py::gil_scoped_release>());

  py::class_<coord::InProcCoord, coord::CoordService,
             std::shared_ptr<coord::InProcCoord>>(m, "InProcCoord")
      .def(py::init([] {
        return std::make_shared<coord::InProcCoord>(
            std::make_shared<coord::CoordStore>());
      }))
      .def(py::init([](std::shared_ptr<coord::CoordStore> store) {
        return std::make_shared<coord::InProcCoord>(std::move(store));
      }))
      .def("store", &coord::InProcCoord::store);

  py::class_<coord::CoordServer>(m, "CoordServer")
      .def(py::init([](std::shared_ptr<coord::CoordStore> store) {
             return std::make_unique<coord::CoordServer>(std::move(store));
           }))
      .def(py::init([] {
        return std::make_unique<coord::CoordServer>(
            std::make_shared<coord::CoordStore>());
      }))
      .def("start", [](coord::CoordServer& s, const std::string& host, uint16_t port) {
        unwrap_void(s.start(host, port));
      }, py::arg("host") = "127.0.0.1", py::arg("port") = 0)
      .def("stop", &coord::CoordServer::stop, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("port", &coord::CoordServer::port)
      .def_property_readonly("endpoint", &coord::CoordServer::endpoint)
      .def("store", &coord::CoordServer::store)
      .def("set_read_only", &coord::CoordServer::set_read_only)
      .def("read_only", &coord::CoordServer::read_only);

  py::class_<coord::CoordFollower>(m, "CoordFollower")
      .def(py::init([](coord::CoordServer& server,
                       const std::string& primary, uint64_t failover_ms) {
             return std::make_unique<coord::CoordFollower>(
                 server.store(), &server, primary, failover_ms);
           }),
           py::arg("server"), py::arg("primary"),
           py::arg("failover_ms") = 2000, py::keep_alive<1, 2>())
      .def("start", [](coord::CoordFollower& f) { unwrap_void(f.start()); },
           py::call_guard<py::gil_scoped_release>())
      .def("stop", &coord::CoordFollower::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("promoted", &coord::CoordFollower::promoted);

  py::class_<coord::CoordClient, coord::CoordService,
             std::shared_ptr<coord::CoordClient>>(m, "CoordClient")
      .def(py::init<>())
      .def("connect", [](coord::CoordClient& c, const std::string& ep) {
        unwrap_void(c.connect(ep));
      }, py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("observed_epoch",
                             &coord::CoordClient::observed_epoch)
      .def("put_many",
           [](coord::CoordClient& c,
              const std::vector<std::pair<std::string, std::string>>& puts,
              const std::vector<std::string>& dels) {
             std::vector<coord::KV> kvs;
             kvs.reserve(puts.size());
             for (auto& [k, v] : puts) kvs.push_back({k, v});
             py::gil_scoped_release rel;
             unwrap_void(c.put_many(kvs, dels));
           },
           py::arg("puts"), py::arg("dels") = std::vector<std::string>{})
      .def("close", &coord::CoordClient::close,
           py::call_guard<py::gil_scoped_release>());

  py::class_<coord::LeaderElector>(m, "LeaderElector")
      .def(py::init<std::shared_ptr<coord::CoordService>, std::string,
                    std::string, uint64_t>(),
           py::arg("coord"), py::arg("key"), py::arg("candidate_id"),
           py::arg("lease_ms") = 5000)
      .def("start", &coord::LeaderElector::start)
      .def("stop", &coord::LeaderElector::stop,
           py::call_guard<py::gil_scoped_release>())
      .def_property_readonly("is_leader", &coord::LeaderElector::is_leader)
      .def("current_leader", &coord::LeaderElector::current_leader);

  // Test-only: serialize a one-shard DATA_PULL request exactly as the wire
  // sees it (hostile-input tests craft out-of-range pulls without mirroring
  // the serde layout in Python).
  m.def("encode_pull_req_for_test",
        [](const std::string& dst_pool, uint64_t dst_offset, uint64_t total_len,
           const std::string& src_pool, uint64_t src_offset, uint64_t src_len) {
          ShardPlacement sp;
          sp.pool_id = src_pool;
          sp.offset = src_offset;
          sp.length = src_len;
          serde::Enc e;
          serde::put(e, dst_pool);
          serde::put(e, dst_offset);
          serde::put(e, total_len);
          serde::put(e, std::vector<ShardPlacement>{sp});
          return py::bytes(e.buf);
        });

  // ------------------------------------------------------------- gpu
  auto gm = m.def_submodule("gpu");
  gm.def("available", &gpu::available);
  gm.def("device_count", &gpu::device_count);
  gm.def("checksum_cpu_tiles", [](py::buffer b, uint64_t first_tile) {
    py::buffer_info info = b.request();
    return gpu::checksum_cpu_tiles(info.ptr,
                                   static_cast<uint64_t>(info.size * info.itemsize),
                                   first_tile);
  });
  gm.def("checksum_cpu_finalize", &gpu::checksum_cpu_finalize);
  gm.def("checksum_cpu", [](py::buffer b) {
    py::buffer_info info = b.request();
    return gpu::checksum_cpu(info.ptr,
                             static_cast<uint64_t>(info.size * info.itemsize));
  });
  gm.def("sync", [] {
    py::gil_scoped_release rel;
    unwrap_void(gpu::sync());
  });
  gm.def("malloc", [](uint64_t nbytes, int device) {
    return unwrap(gpu::device_malloc(nbytes, device));
  }, py::arg("nbytes"), py::arg("device") = 0);
  gm.def("free", [](uint64_t ptr) { unwrap_void(gpu::device_free(ptr)); });
  gm.def("upload", [](uint64_t dst, py::buffer b) {
    py::buffer_info info = b.request();
    unwrap_void(gpu::upload(dst, info.ptr,
                            static_cast<uint64_t>(info.size * info.itemsize)));
  });
  gm.def("download", [](uint64_t src, uint64_t nbytes) {
    std::string out;
    out.resize(nbytes);
    unwrap_void(gpu::download(out.data(), src, nbytes));
    return py::bytes(out);
  });
  gm.def("checksum_device", [](uint64_t ptr, uint64_t nbytes, int device) {
    py::gil_scoped_release rel;
    return unwrap(gpu::checksum_sync(reinterpret_cast<const void*>(ptr), nbytes,
                                     device, nullptr));
  }, py::arg("ptr"), py::arg("nbytes"), py::arg("device") = 0);
  gm.def("checksum_device_batch",
         [](const std::vector<std::pair<uint64_t, uint64_t>>& objs, int device) {
           std::vector<const void*> ptrs;
           std::vector<uint64_t> sizes;
           for (auto& [p, s] : objs) {
             ptrs.push_back(reinterpret_cast<const void*>(p));
             sizes.push_back(s);
           }
           std::vector<uint64_t> out(objs.size());
           py::gil_scoped_release rel;
           unwrap_void(gpu::checksum_batch(ptrs.data(), sizes.data(),
                                           static_cast<uint32_t>(objs.size()),
                                           out.data(), device, nullptr));
           return out;
         }, py::arg("objs"), py::arg("device") = 0);
  gm.def("batched_copy",
         [](const std::vector<std::tuple<uint64_t, uint64_t, uint64_t>>& descs) {
           std::vector<gpu::CopyDesc> ds;
           for (auto& [src, dst, n] : descs)
             ds.push_back({reinterpret_cast<const void*>(src),
                           reinterpret_cast<void*>(dst), n});
           py::gil_scoped_release rel;
           unwrap_void(gpu::batched_copy(ds.data(),
                                         static_cast<uint32_t>(ds.size()), nullptr));
           unwrap_void(gpu::sync());
         });
  gm.def("fill_pattern", [](uint64_t ptr, uint64_t nbytes, uint64_t seed) {
    py::gil_scoped_release rel;
    unwrap_void(gpu::fill_pattern(reinterpret_cast<void*>(ptr), nbytes, seed, nullptr));
    unwrap_void(gpu::sync());
  }, py::arg("ptr"), py::arg("nbytes"), py::arg("seed") = 0);
  gm.def("verify_pattern", [](uint64_t ptr, uint64_t nbytes, uint64_t seed) {
    py::gil_scoped_release rel;
    return unwrap(gpu::verify_pattern(reinterpret_cast<const void*>(ptr), nbytes,
                                      seed, nullptr));
  }, py::arg("ptr"), py::arg("nbytes"), py::arg("seed") = 0);
  gm.def("mfma_i8_probe", [](py::buffer a, py::buffer b, int device) {
    py::buffer_info ia = a.request(), ib = b.request();
    if (ia.size * ia.itemsize != 1024 || ib.size * ib.itemsize != 1024)
      throw std::runtime_error("A and B must be 1024 bytes (32x32 i8)");
    std::vector<int32_t> c(1024);
    unwrap_void(gpu::mfma_i8_probe(static_cast<const int8_t*>(ia.ptr),
                                   static_cast<const int8_t*>(ib.ptr), c.data(),
                                   device));
    return c;
  }, py::arg("a"), py::arg("b"), py::arg("device") = 0);

  bind_store(m);
}
