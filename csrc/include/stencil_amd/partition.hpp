// Hierarchical 3D domain partition (node level x GPU level).
//
// MI355X-native equivalent of the reference's RankPartition/NodePartition
// (reference: include/stencil/partition.hpp:20-256). Splits the global grid
// recursively by the prime factors of the node/GPU counts, each time along
// the axis whose radius-weighted interface area is smallest; uneven
// remainders give the first (total % dim) subdomains one extra element.
#pragma once

#include <vector>

#include "stencil_amd/core.hpp"

namespace stencil_amd {

namespace detail {
// Greedy split: repeatedly divide `size` by the prime factors of n, choosing
// per factor the axis that minimizes the radius-weighted interface created.
inline void split_by_iface(Vec3 &size, Vec3 &dim, int64_t n, const Radius &radius) {
  for (int64_t amt : prime_factors(n)) {
    if (amt < 2) continue;
    const int64_t xI = size.y * size.z * (radius.dir(1, 0, 0) + radius.dir(-1, 0, 0));
    const int64_t yI = size.x * size.z * (radius.dir(0, 1, 0) + radius.dir(0, -1, 0));
    const int64_t zI = size.x * size.y * (radius.dir(0, 0, 1) + radius.dir(0, 0, -1));
    if (xI <= yI && xI <= zI) {
      size.x = div_ceil(size.x, amt);
      dim.x *= amt;
    } else if (yI <= zI) {
      size.y = div_ceil(size.y, amt);
      dim.y *= amt;
    } else {
      size.z = div_ceil(size.z, amt);
      dim.z *= amt;
    }
  }
}
// Split preferring the LARGEST axis (radius-agnostic; the reference's
// RankPartition tie-breaking, partition.hpp:36-47).
inline void split_by_largest(Vec3 &size, Vec3 &dim, int64_t n) {
  for (int64_t amt : prime_factors(n)) {
    if (amt < 2) continue;
    if (size.x >= size.y && size.x >= size.z) {
      size.x = div_ceil(size.x, amt);
      dim.x *= amt;
    } else if (size.y >= size.z) {
      size.y = div_ceil(size.y, amt);
      dim.y *= amt;
    } else {
      size.z = div_ceil(size.z, amt);
      dim.z *= amt;
    }
  }
}

inline int64_t linearize(const Vec3 &idx, const Vec3 &dim) {
  assert(idx.all_ge(0) && idx.x < dim.x && idx.y < dim.y && idx.z < dim.z);
  return idx.x + idx.y * dim.x + idx.z * dim.y * dim.x;
}
inline Vec3 dimensionize(int64_t i, const Vec3 &dim) {
  assert(i >= 0 && i < dim.flatten());
  Vec3 ret;
  ret.x = i % dim.x;
  i /= dim.x;
  ret.y = i % dim.y;
  i /= dim.y;
  ret.z = i;
  return ret;
}
} // namespace detail

// Flat partition of `size` into n subdomains, splitting the largest axis first.
class RankPartition {
  Vec3 dim_{1, 1, 1};
  Vec3 size_; // ceil size of a subdomain
  Vec3 rem_;  // total % dim : number of "big" subdomains per axis

public:
  RankPartition(const Vec3 &size, int64_t n) : size_(size) {
    detail::split_by_largest(size_, dim_, n);
    rem_ = size % dim_;
  }

  Vec3 dim() const { return dim_; }

  Vec3 subdomain_size(const Vec3 &idx) const {
    Vec3 ret = size_;
    if (rem_.x != 0 && idx.x >= rem_.x) ret.x -= 1;
    if (rem_.y != 0 && idx.y >= rem_.y) ret.y -= 1;
    if (rem_.z != 0 && idx.z >= rem_.z) ret.z -= 1;
    return ret;
  }

  Vec3 subdomain_origin(const Vec3 &idx) const {
    Vec3 ret = size_ * idx;
    if (rem_.x != 0 && idx.x >= rem_.x) ret.x -= (idx.x - rem_.x);
    if (rem_.y != 0 && idx.y >= rem_.y) ret.y -= (idx.y - rem_.y);
    if (rem_.z != 0 && idx.z >= rem_.z) ret.z -= (idx.z - rem_.z);
    return ret;
  }

  int64_t linearize(const Vec3 &idx) const { return detail::linearize(idx, dim()); }
  Vec3 dimensionize(int64_t i) const { return detail::dimensionize(i, dim()); }
};

// Two-level partition: system (nodes) x node (GPUs), minimizing
// radius-weighted interface area at each split.
class NodePartition {
  Vec3 sysDim_{1, 1, 1};
  Vec3 nodeDim_{1, 1, 1};
  Vec3 size_;
  Vec3 rem_;

public:
  NodePartition() : NodePartition(Vec3(1, 1, 1), Radius::constant(0), 1, 1) {}
  NodePartition(const Vec3 &size, const Radius &radius, int64_t nodes, int64_t gpus) : size_(size) {
    detail::split_by_iface(size_, sysDim_, nodes, radius);
    detail::split_by_iface(size_, nodeDim_, gpus, radius);
    rem_ = size % (sysDim_ * nodeDim_);
  }

  Vec3 sys_dim() const { return sysDim_; }
  Vec3 node_dim() const { return nodeDim_; }
  Vec3 dim() const { return sysDim_ * nodeDim_; }

  Vec3 subdomain_size(const Vec3 &idx) const {
    Vec3 ret = size_;
    if (rem_.x != 0 && idx.x >= rem_.x) ret.x -= 1;
    if (rem_.y != 0 && idx.y >= rem_.y) ret.y -= 1;
    if (rem_.z != 0 && idx.z >= rem_.z) ret.z -= 1;
    return ret;
  }

  Vec3 subdomain_origin(const Vec3 &idx) const {
    Vec3 ret = size_ * idx;
    if (rem_.x != 0 && idx.x >= rem_.x) ret.x -= (idx.x - rem_.x);
    if (rem_.y != 0 && idx.y >= rem_.y) ret.y -= (idx.y - rem_.y);
    if (rem_.z != 0 && idx.z >= rem_.z) ret.z -= (idx.z - rem_.z);
    return ret;
  }

  Vec3 sys_idx(int64_t i) const { return detail::dimensionize(i, sys_dim()); }
  Vec3 node_idx(int64_t i) const { return detail::dimensionize(i, node_dim()); }
};

} // namespace stencil_amd
