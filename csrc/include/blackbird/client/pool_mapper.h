// Cached one-sided pool mappings shared by the host Client and GpuClient:
// POSIX-shm segments (host tier) and hipIpc-imported HBM pools (GPU tier).
#pragma once

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <map>
#include <mutex>
#include <set>
#include <string>
#include <vector>

#include <hip/hip_runtime_api.h>

#include "blackbird/common/hex.h"

namespace blackbird {

class PoolMapper {
 public:
  ~PoolMapper() {
    for (void* p : registered_) (void)hipHostUnregister(p);
    for (auto& [k, m] : shm_) munmap(m.ptr, m.size);
    for (auto& [k, p] : ipc_) (void)hipIpcCloseMemHandle(p);
  }

  // SHM: returns mapped base or nullptr (maps the whole segment; size taken
  // from the segment itself, reported via *mapped_size when non-null).
  void* map_shm(const std::string& name, uint64_t size_hint,
                uint64_t* mapped_size = nullptr) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = shm_.find(name);
    if (it != shm_.end()) {
      if (mapped_size) *mapped_size = it->second.size;
      return it->second.ptr;
    }
    int fd = shm_open(name.c_str(), O_RDWR, 0600);
    if (fd < 0) return nullptr;
    struct stat st {};
    uint64_t size = size_hint;
    if (fstat(fd, &st) == 0 && st.st_size > 0)
      size = static_cast<uint64_t>(st.st_size);
    if (size == 0) {
      ::close(fd);
      return nullptr;
    }
    void* p = mmap(nullptr, size, PROT_READ | PROT_WRITE, MAP_SHARED, fd, 0);
    ::close(fd);
    if (p == MAP_FAILED) return nullptr;
    shm_[name] = {p, size};
    if (mapped_size) *mapped_size = size;
    return p;
  }

  // Device-visible mapping of a HOST pool: pins (hipHostRegister, unless the
  // memory is already page-locked) and maps it into the GPU address space
  // once, cached per pool. Lets the fused copy/digest kernels read and write
  // host tiers (PINNED_CPU, local RAM_CPU/shm) directly at PCIe DMA speed
  // instead of double-copying through a staging bounce. nullptr when the GPU
  // cannot address it (registration failed / no GPU).
  void* host_dev_map(const std::string& pool_key, void* base, uint64_t size) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = dev_maps_.find(pool_key);
    if (it != dev_maps_.end()) return it->second;
    if (dev_map_failed_.count(pool_key)) return nullptr;
    void* dp = nullptr;
    if (hipHostGetDevicePointer(&dp, base, 0) != hipSuccess) {
      // the probe failing is EXPECTED for not-yet-registered memory — clear
      // the sticky thread error so a later launch check doesn't inherit it
      (void)hipGetLastError();
      if (hipHostRegister(base, size, hipHostRegisterDefault) != hipSuccess) {
        (void)hipGetLastError();
        dev_map_failed_.insert(pool_key);
        return nullptr;
      }
      registered_.push_back(base);
      if (hipHostGetDevicePointer(&dp, base, 0) != hipSuccess) {
        (void)hipGetLastError();
        dev_map_failed_.insert(pool_key);
        return nullptr;
      }
    }
    dev_maps_[pool_key] = dp;
    return dp;
  }

  // HIP IPC: returns device pointer valid in this process, or nullptr.
  // Negative cache: a handle that failed to open is not retried (handles are
  // immutable per allocation).
  void* open_ipc(const std::string& handle_hex, int device) {
    std::lock_guard<std::mutex> g(mu_);
    auto it = ipc_.find(handle_hex);
    if (it != ipc_.end()) return it->second;
    if (ipc_failed_.count(handle_hex)) return nullptr;
    hipIpcMemHandle_t h{};
    if (!from_hex(handle_hex, &h, sizeof(h))) return nullptr;
    void* p = nullptr;
    if (hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess) != hipSuccess) {
      (void)hipGetLastError();  // don't leak the probe error to launch checks
      ipc_failed_.insert(handle_hex);
      return nullptr;
    }
    ipc_[handle_hex] = p;
    return p;
  }

 private:
  struct Shm {
    void* ptr;
    uint64_t size;
  };
  std::mutex mu_;
  std::map<std::string, Shm> shm_;
  std::map<std::string, void*> ipc_;
  std::set<std::string> ipc_failed_;
  std::map<std::string, void*> dev_maps_;  // pool key → device-visible ptr
  std::set<std::string> dev_map_failed_;
  std::vector<void*> registered_;  // bases WE pinned (unregister on teardown)
};

}  // namespace blackbird
