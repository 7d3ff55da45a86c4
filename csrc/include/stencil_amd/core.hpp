// stencil_amd core geometry types: Vec3, Rect3, DirectionMap, Radius.
//
// MI355X-native re-design of the foundation layer of cwpearson/stencil
// (reference: include/stencil/dim3.hpp, rect3.hpp, radius.hpp,
// direction_map.hpp). Fixes the reference's Dim3 operator!=/max bugs
// (dim3.hpp:195, :57-63) rather than replicating them.
#pragma once

#include <algorithm>
#include <array>
#include <cassert>
#include <vector>
#include <cstdint>
#include <functional>
#include <ostream>
#include <string>

namespace stencil_amd {

struct Vec3 {
  int64_t x = 0, y = 0, z = 0;

  constexpr Vec3() = default;
  constexpr Vec3(int64_t x_, int64_t y_, int64_t z_) : x(x_), y(y_), z(z_) {}

  constexpr int64_t &operator[](int i) { return i == 0 ? x : (i == 1 ? y : z); }
  constexpr const int64_t &operator[](int i) const { return i == 0 ? x : (i == 1 ? y : z); }

  constexpr Vec3 operator+(const Vec3 &o) const { return {x + o.x, y + o.y, z + o.z}; }
  constexpr Vec3 operator-(const Vec3 &o) const { return {x - o.x, y - o.y, z - o.z}; }
  constexpr Vec3 operator*(const Vec3 &o) const { return {x * o.x, y * o.y, z * o.z}; }
  constexpr Vec3 operator/(const Vec3 &o) const { return {x / o.x, y / o.y, z / o.z}; }
  constexpr Vec3 operator%(const Vec3 &o) const { return {x % o.x, y % o.y, z % o.z}; }
  constexpr Vec3 operator*(int64_t s) const { return {x * s, y * s, z * s}; }
  constexpr Vec3 operator-() const { return {-x, -y, -z}; }
  Vec3 &operator+=(const Vec3 &o) { x += o.x; y += o.y; z += o.z; return *this; }
  Vec3 &operator-=(const Vec3 &o) { x -= o.x; y -= o.y; z -= o.z; return *this; }

  constexpr bool operator==(const Vec3 &o) const { return x == o.x && y == o.y && z == o.z; }
  constexpr bool operator!=(const Vec3 &o) const { return !(*this == o); }
  // lexicographic (z, y, x) so it can key std::map like the reference's Dim3
  constexpr bool operator<(const Vec3 &o) const {
    if (z != o.z) return z < o.z;
    if (y != o.y) return y < o.y;
    return x < o.x;
  }

  constexpr int64_t flatten() const { return x * y * z; }
  constexpr bool any_eq(int64_t v) const { return x == v || y == v || z == v; }
  constexpr bool all_ge(int64_t v) const { return x >= v && y >= v && z >= v; }
  constexpr bool all_le(int64_t v) const { return x <= v && y <= v && z <= v; }

  // component-wise max/min (the reference's Dim3::max used x for all three)
  static constexpr Vec3 max(const Vec3 &a, const Vec3 &b) {
    return {a.x > b.x ? a.x : b.x, a.y > b.y ? a.y : b.y, a.z > b.z ? a.z : b.z};
  }
  static constexpr Vec3 min(const Vec3 &a, const Vec3 &b) {
    return {a.x < b.x ? a.x : b.x, a.y < b.y ? a.y : b.y, a.z < b.z ? a.z : b.z};
  }

  // wrap into [0, extent) per axis (periodic boundary)
  Vec3 wrap(const Vec3 &extent) const {
    auto w = [](int64_t v, int64_t e) {
      v %= e;
      return v < 0 ? v + e : v;
    };
    return {w(x, extent.x), w(y, extent.y), w(z, extent.z)};
  }

  std::string str() const {
    return "[" + std::to_string(x) + "," + std::to_string(y) + "," + std::to_string(z) + "]";
  }
};

inline std::ostream &operator<<(std::ostream &os, const Vec3 &v) { return os << v.str(); }

// half-open box [lo, hi)
struct Rect3 {
  Vec3 lo, hi;
  constexpr Rect3() = default;
  constexpr Rect3(const Vec3 &l, const Vec3 &h) : lo(l), hi(h) {}
  constexpr Vec3 extent() const { return hi - lo; }
  constexpr int64_t volume() const {
    Vec3 e = extent();
    return (e.x > 0 && e.y > 0 && e.z > 0) ? e.flatten() : 0;
  }
  constexpr bool contains(const Vec3 &p) const {
    return p.x >= lo.x && p.x < hi.x && p.y >= lo.y && p.y < hi.y && p.z >= lo.z && p.z < hi.z;
  }
  constexpr bool operator==(const Rect3 &o) const { return lo == o.lo && hi == o.hi; }
  std::string str() const { return lo.str() + "..." + hi.str(); }
};

inline std::ostream &operator<<(std::ostream &os, const Rect3 &r) { return os << r.str(); }

// Map from a 3D direction vector (components in {-1,0,1}) to a value.
template <typename T> class DirectionMap {
  std::array<T, 27> v_{};

public:
  T &at_dir(int x, int y, int z) {
    assert(x >= -1 && x <= 1 && y >= -1 && y <= 1 && z >= -1 && z <= 1);
    return v_[(z + 1) * 9 + (y + 1) * 3 + (x + 1)];
  }
  const T &at_dir(int x, int y, int z) const {
    assert(x >= -1 && x <= 1 && y >= -1 && y <= 1 && z >= -1 && z <= 1);
    return v_[(z + 1) * 9 + (y + 1) * 3 + (x + 1)];
  }
  bool operator==(const DirectionMap &o) const { return v_ == o.v_; }
};

// Per-direction stencil radius (reference: include/stencil/radius.hpp).
class Radius {
  DirectionMap<int64_t> rads_;

public:
  int64_t &dir(int x, int y, int z) { return rads_.at_dir(x, y, z); }
  const int64_t &dir(int x, int y, int z) const { return rads_.at_dir(x, y, z); }
  int64_t &dir(const Vec3 &d) { return dir((int)d.x, (int)d.y, (int)d.z); }
  const int64_t &dir(const Vec3 &d) const { return dir((int)d.x, (int)d.y, (int)d.z); }

  int64_t x(int d) const { return dir(d, 0, 0); }
  int64_t y(int d) const { return dir(0, d, 0); }
  int64_t z(int d) const { return dir(0, 0, d); }

  bool operator==(const Radius &o) const { return rads_ == o.rads_; }

  static Radius constant(int64_t r) {
    Radius ret;
    for (int z = -1; z <= 1; ++z)
      for (int y = -1; y <= 1; ++y)
        for (int x = -1; x <= 1; ++x)
          ret.dir(x, y, z) = r;
    ret.dir(0, 0, 0) = 0;
    return ret;
  }

  // distinct radii for faces / edges / corners
  static Radius face_edge_corner(int64_t face, int64_t edge, int64_t corner) {
    Radius ret;
    for (int z = -1; z <= 1; ++z)
      for (int y = -1; y <= 1; ++y)
        for (int x = -1; x <= 1; ++x) {
          int nz = (x != 0) + (y != 0) + (z != 0);
          int64_t r = nz == 1 ? face : (nz == 2 ? edge : (nz == 3 ? corner : 0));
          ret.dir(x, y, z) = r;
        }
    return ret;
  }
};

// numeric helpers (reference: include/stencil/numeric.hpp)
inline int64_t div_ceil(int64_t n, int64_t d) { return (n + d - 1) / d; }

// prime factors of n, sorted descending
inline std::vector<int64_t> prime_factors(int64_t n) {
  std::vector<int64_t> out;
  for (int64_t p = 2; p * p <= n; ++p)
    while (n % p == 0) {
      out.push_back(p);
      n /= p;
    }
  if (n > 1) out.push_back(n);
  std::sort(out.rbegin(), out.rend());
  return out;
}

inline int64_t align_up(int64_t x, int64_t a) { return (x + a - 1) / a * a; }

} // namespace stencil_amd
