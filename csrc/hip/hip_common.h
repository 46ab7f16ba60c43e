// Shared helpers for the gfx950 kernel TUs.
#pragma once

#include <hip/hip_runtime.h>

#include <string>

#include "blackbird/common/result.h"

namespace blackbird::gpu {

inline Error hip_error(hipError_t e, const char* what) {
  return Error{ErrorCode::HIP_ERROR,
               std::string(what) + ": " + hipGetErrorString(e)};
}

#define BB_HIP_TRY(expr)                                        \
  do {                                                          \
    hipError_t _e = (expr);                                     \
    if (_e != hipSuccess) {                                     \
      (void)hipGetLastError(); /* consume sticky thread error */ \
      return ::blackbird::gpu::hip_error(_e, #expr);            \
    }                                                           \
  } while (0)

}  // namespace blackbird::gpu
