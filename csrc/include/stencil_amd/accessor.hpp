// Grid-coordinate accessor for C++ app kernels (reference:
// include/stencil/accessor.hpp, pitched_ptr.hpp): index a pitched
// allocation by GLOBAL 3D coordinates; the origin already accounts for the
// halo offset, so acc[Vec3(x,y,z)] is valid over the full region
// (compute region + halos).
#pragma once

#include "stencil_amd/core.hpp"
#include "stencil_amd/domain.hpp"

namespace stencil_amd {

template <typename T> struct Accessor {
  char *ptr = nullptr;   // allocation base
  int64_t pitch = 0;     // bytes between y rows
  int64_t plane = 0;     // bytes between z planes
  Vec3 origin;           // global coordinate of allocation element (0,0,0)

  Accessor() = default;
  Accessor(const Pitched &p, const Vec3 &allocOrigin)
      : ptr(p.ptr), pitch(p.pitch), plane(p.plane()), origin(allocOrigin) {}

  __host__ __device__ T &operator[](const Vec3 &g) const {
    return *reinterpret_cast<T *>(ptr + (g.z - origin.z) * plane + (g.y - origin.y) * pitch +
                                  (g.x - origin.x) * (int64_t)sizeof(T));
  }
};

// accessor over the CURRENT buffer of quantity qi of a realized domain
template <typename T> Accessor<T> curr_accessor(const LocalDomain &d, int64_t qi) {
  return Accessor<T>(d.curr(qi), d.full_region().lo);
}
template <typename T> Accessor<T> next_accessor(const LocalDomain &d, int64_t qi) {
  return Accessor<T>(d.next(qi), d.full_region().lo);
}

} // namespace stencil_amd
