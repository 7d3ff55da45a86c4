// Device-side helpers.
#pragma once

#include <hip/hip_runtime.h>

namespace stencil_amd {

// Make a pointer loaded from memory provably wave-uniform so the compiler
// keeps it in SGPRs and emits scalar-base addressing for every dependent
// load/store (instead of 64-bit per-lane VGPR address math). The slot
// values ARE uniform (same slot for the whole grid); the compiler just
// cannot prove it.
template <typename T> __device__ __forceinline__ T *uniform_ptr(T *p) {
  const uint64_t v = (uint64_t)p;
  const uint32_t lo = __builtin_amdgcn_readfirstlane((uint32_t)v);
  const uint32_t hi = __builtin_amdgcn_readfirstlane((uint32_t)(v >> 32));
  return (T *)(((uint64_t)hi << 32) | lo);
}

} // namespace stencil_amd
