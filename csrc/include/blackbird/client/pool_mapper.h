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

#include <hip/hip_runtime_api.h>

#include "blackbird/common/hex.h"

namespace blackbird {

class PoolMapper {
 public:
  ~PoolMapper() {
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
};

}  // namespace blackbird
