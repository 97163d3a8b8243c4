// Shared HIP helpers for the MI355X backend.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdlib>
#include <stdexcept>
#include <string>

namespace rga::hip {

#define RGA_HIP_CHECK(expr)                                                          \
  do {                                                                               \
    hipError_t rga_hip_err_ = (expr);                                                \
    if (rga_hip_err_ != hipSuccess) {                                                \
      fprintf(stderr, "[rga::hip] error: %s failed: %s (%s:%d)\n", #expr,            \
              hipGetErrorString(rga_hip_err_), __FILE__, __LINE__);                  \
      exit(1);                                                                       \
    }                                                                                \
  } while (0)

// Throwing variant for recoverable paths (arena construction): callers fall
// back to smaller arenas or the CPU engine instead of dying.
#define RGA_HIP_TRY(expr)                                                            \
  do {                                                                               \
    hipError_t rga_hip_err_ = (expr);                                                \
    if (rga_hip_err_ != hipSuccess) {                                                \
      throw std::runtime_error(std::string("[rga::hip] ") + #expr + " failed: " +    \
                               hipGetErrorString(rga_hip_err_));                     \
    }                                                                                \
  } while (0)

inline int device_count() {
  int n = 0;
  hipError_t err = hipGetDeviceCount(&n);
  if (err != hipSuccess) {
    return 0;
  }
  return n;
}

}  // namespace rga::hip
