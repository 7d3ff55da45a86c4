#pragma once

#include <hip/hip_runtime.h>

#include <sstream>
#include <stdexcept>
#include <string>

namespace stencil_amd {

inline void hip_check(hipError_t err, const char *file, int line) {
  if (err != hipSuccess) {
    std::ostringstream ss;
    ss << "HIP error at " << file << ":" << line << ": " << hipGetErrorString(err);
    throw std::runtime_error(ss.str());
  }
}

#define STENCIL_HIP(expr) ::stencil_amd::hip_check((expr), __FILE__, __LINE__)

} // namespace stencil_amd
