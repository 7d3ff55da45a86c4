// Subdomain -> (rank, GPU) placement strategies, C++ side.
//
// Same semantics as the Python implementation (stencil_amd/parallel/
// placement.py) over the same NodePartition + QAP pieces, so the C++
// orchestrator (distributed.hpp) and the Python DistributedDomain compute
// IDENTICAL assignments — the parity is pinned by tests/test_native_plan.py.
// MI355X-native equivalent of the reference's Placement hierarchy
// (reference: include/stencil/partition.hpp:258-831).
//
// IntraNodeRandom uses std::mt19937 where Python uses random.Random: the
// two RNGs differ by design (it is the experimental control, not a wire
// format), so cross-language parity is only guaranteed for Trivial and
// NodeAware.
#pragma once

#include <algorithm>
#include <map>
#include <numeric>
#include <random>
#include <set>
#include <stdexcept>
#include <utility>
#include <vector>

#include <hip/hip_runtime.h>

#include "stencil_amd/core.hpp"
#include "stencil_amd/partition.hpp"
#include "stencil_amd/qap.hpp"
#include "stencil_amd/topo.hpp"

namespace stencil_amd {

enum class PlacementStrategy { NodeAware, Trivial, IntraNodeRandom };

// one (rank, local domain index, HIP device, node) GPU slot
struct Slot {
  int rank;
  int localId;
  int cuda;
  int node;
};

class Placement {
public:
  Placement(const Vec3 &size, const Radius &radius, std::vector<Slot> slots)
      : radius_(radius), slots_(std::move(slots)) {
    for (const Slot &s : slots_) numLocal_[s.rank] += 1;
    std::map<int, int64_t> perNode;
    for (const Slot &s : slots_) perNode[s.node] += 1;
    const int64_t nNodes = (int64_t)perNode.size();
    std::set<int64_t> counts;
    for (auto &kv : perNode) counts.insert(kv.second);
    uniformNodes_ = counts.size() <= 1;
    if (uniformNodes_) {
      part_ = NodePartition(size, radius, nNodes, perNode.empty() ? 1 : perNode.begin()->second);
    } else {
      // heterogeneous per-node GPU counts: flat single-level split
      part_ = NodePartition(size, radius, 1, (int64_t)slots_.size());
    }
  }
  virtual ~Placement() = default;

  Vec3 dim() const { return part_.dim(); }
  int64_t linearize(const Vec3 &idx) const { return detail::linearize(idx, dim()); }
  Vec3 dimensionize(int64_t gid) const { return detail::dimensionize(gid, dim()); }

  int get_rank(const Vec3 &idx) const { return slots_[assign_.at(linearize(idx))].rank; }
  int get_subdomain_id(const Vec3 &idx) const {
    return slots_[assign_.at(linearize(idx))].localId;
  }
  int get_cuda(const Vec3 &idx) const { return slots_[assign_.at(linearize(idx))].cuda; }
  Vec3 get_idx(int rank, int localId) const { return byRank_.at({rank, localId}); }
  int num_local(int rank) const {
    auto it = numLocal_.find(rank);
    return it == numLocal_.end() ? 0 : (int)it->second;
  }
  Vec3 subdomain_size(const Vec3 &idx) const { return part_.subdomain_size(idx); }
  Vec3 subdomain_origin(const Vec3 &idx) const { return part_.subdomain_origin(idx); }
  const std::vector<Slot> &slots() const { return slots_; }
  bool uniform_nodes() const { return uniformNodes_; }

protected:
  void finish_() {
    for (auto &kv : assign_) {
      const Slot &s = slots_[kv.second];
      byRank_[{s.rank, s.localId}] = dimensionize(kv.first);
    }
  }

  // node n owns the sys-block of subdomains (same enumeration order as
  // the Python _node_gids: x fastest)
  std::vector<int64_t> node_gids_(int node) const {
    const Vec3 sd = part_.sys_dim(), nd = part_.node_dim();
    const int64_t sx = node % sd.x, sy = (node / sd.x) % sd.y, sz = node / (sd.x * sd.y);
    std::vector<int64_t> gids;
    for (int64_t z = 0; z < nd.z; ++z)
      for (int64_t y = 0; y < nd.y; ++y)
        for (int64_t x = 0; x < nd.x; ++x)
          gids.push_back(linearize(Vec3(sx * nd.x + x, sy * nd.y + y, sz * nd.z + z)));
    return gids;
  }

  void check_counts_() const {
    const int64_t n = dim().flatten();
    if (n != (int64_t)slots_.size())
      throw std::runtime_error("placement: subdomain/slot count mismatch");
  }

  int n_nodes_() const {
    std::set<int> nodes;
    for (const Slot &s : slots_) nodes.insert(s.node);
    return (int)nodes.size();
  }

  Radius radius_;
  std::vector<Slot> slots_;
  NodePartition part_;
  bool uniformNodes_ = true;
  std::map<int64_t, int64_t> assign_; // gid -> slot index
  std::map<std::pair<int, int>, Vec3> byRank_;
  std::map<int, int64_t> numLocal_;
};

class TrivialPlacement : public Placement {
public:
  TrivialPlacement(const Vec3 &size, const Radius &radius, std::vector<Slot> slots)
      : Placement(size, radius, std::move(slots)) {
    check_counts_();
    const int64_t n = dim().flatten();
    for (int64_t gid = 0; gid < n; ++gid) assign_[gid] = gid;
    finish_();
  }
};

class NodeAwarePlacement : public Placement {
public:
  // haloExtent(negDir, size, radius) supplied by the caller to avoid a
  // domain.hpp dependency cycle; distributed.hpp passes
  // LocalDomain::halo_extent
  using HaloExtentFn = Vec3 (*)(const Vec3 &, const Vec3 &, const Radius &);

  NodeAwarePlacement(const Vec3 &size, const Radius &radius, std::vector<Slot> slots,
                     HaloExtentFn haloExtent, double qapTimeoutSec = 10.0)
      : Placement(size, radius, std::move(slots)) {
    check_counts_();
    const int64_t n = dim().flatten();
    if (!uniformNodes_) { // no node blocking to optimize within
      for (int64_t gid = 0; gid < n; ++gid) assign_[gid] = gid;
      finish_();
      return;
    }
    int haveGpu = 0;
    (void)hipGetDeviceCount(&haveGpu);
    for (int node = 0; node < n_nodes_(); ++node) {
      const std::vector<int64_t> gids = node_gids_(node);
      std::vector<int64_t> slotIds;
      for (size_t i = 0; i < slots_.size(); ++i)
        if (slots_[i].node == node) slotIds.push_back((int64_t)i);
      const SqMat w = comm_matrix_(gids, haloExtent);
      const SqMat d = bandwidth_matrix_(slotIds, haveGpu > 0);
      const std::vector<int64_t> f = qap::solve(w, d, qapTimeoutSec);
      for (size_t a = 0; a < gids.size(); ++a) assign_[gids[a]] = slotIds[f[a]];
    }
    finish_();
  }

private:
  SqMat comm_matrix_(const std::vector<int64_t> &gids, HaloExtentFn haloExtent) const {
    const int64_t n = (int64_t)gids.size();
    SqMat w(n, 0.0);
    const Vec3 d = dim();
    std::map<int64_t, int64_t> indexOf;
    for (int64_t i = 0; i < n; ++i) indexOf[gids[i]] = i;
    for (int64_t gid : gids) {
      const Vec3 p = dimensionize(gid);
      for (int dz = -1; dz <= 1; ++dz)
        for (int dy = -1; dy <= 1; ++dy)
          for (int dx = -1; dx <= 1; ++dx) {
            if (dx == 0 && dy == 0 && dz == 0) continue;
            if (radius_.dir(-dx, -dy, -dz) == 0) continue;
            const Vec3 nb((p.x + dx + d.x) % d.x, (p.y + dy + d.y) % d.y,
                          (p.z + dz + d.z) % d.z);
            const int64_t ngid = linearize(nb);
            auto it = indexOf.find(ngid);
            if (it == indexOf.end() || ngid == gid) continue;
            const Vec3 ext = haloExtent(Vec3(-dx, -dy, -dz), subdomain_size(nb), radius_);
            w.at(indexOf.at(gid), it->second) += (double)(ext.x * ext.y * ext.z);
          }
    }
    return w;
  }

  SqMat bandwidth_matrix_(const std::vector<int64_t> &slotIds, bool haveGpu) const {
    const int64_t n = (int64_t)slotIds.size();
    SqMat d(n, 0.0);
    for (int64_t i = 0; i < n; ++i)
      for (int64_t j = 0; j < n; ++j) {
        const int ci = slots_[slotIds[i]].cuda, cj = slots_[slotIds[j]].cuda;
        d.at(i, j) = haveGpu ? gpu_distance(ci, cj) : (ci == cj ? 0.1 : 1.0);
      }
    return d;
  }
};

class IntraNodeRandomPlacement : public Placement {
public:
  IntraNodeRandomPlacement(const Vec3 &size, const Radius &radius, std::vector<Slot> slots,
                           uint64_t seed = 0)
      : Placement(size, radius, std::move(slots)) {
    check_counts_();
    const int64_t n = dim().flatten();
    if (!uniformNodes_) {
      for (int64_t gid = 0; gid < n; ++gid) assign_[gid] = gid;
      finish_();
      return;
    }
    std::mt19937_64 rng(seed);
    for (int node = 0; node < n_nodes_(); ++node) {
      const std::vector<int64_t> gids = node_gids_(node);
      std::vector<int64_t> slotIds;
      for (size_t i = 0; i < slots_.size(); ++i)
        if (slots_[i].node == node) slotIds.push_back((int64_t)i);
      std::vector<int64_t> perm(slotIds.size());
      std::iota(perm.begin(), perm.end(), 0);
      std::shuffle(perm.begin(), perm.end(), rng);
      for (size_t a = 0; a < gids.size(); ++a) assign_[gids[a]] = slotIds[perm[a]];
    }
    finish_();
  }
};

} // namespace stencil_amd
