// Common helpers for the hipshuffle native library (MI355X / gfx950).
//
// Replaces what libdisni.so + libibverbs did for the reference
// (SURVEY.md §2.3): memory "registration" is hipMalloc + hipIpcGetMemHandle,
// one-sided READ is an xGMI peer copy on a dedicated stream.
#pragma once

#include <hip/hip_runtime.h>

#include <cstdint>
#include <stdexcept>
#include <string>

#define HIP_CHECK(expr)                                                     \
  do {                                                                      \
    hipError_t _e = (expr);                                                 \
    if (_e != hipSuccess) {                                                 \
      throw std::runtime_error(std::string("HIP error at ") + __FILE__ +    \
                               ":" + std::to_string(__LINE__) + ": " +      \
                               hipGetErrorString(_e) + " in " #expr);       \
    }                                                                       \
  } while (0)

namespace hipshuffle {

constexpr int kWave = 64;  // CDNA wavefront width — NOT 32

inline int device_count() {
  int n = 0;
  HIP_CHECK(hipGetDeviceCount(&n));
  return n;
}

}  // namespace hipshuffle
