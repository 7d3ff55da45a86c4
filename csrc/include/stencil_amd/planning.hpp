// Halo-exchange message planning, C++ side (header-only, pure logic).
//
// Exact mirror of stencil_amd/parallel/planning.py — both produce the same
// plan from the same placement (pinned by tests/test_native_plan.py), so a
// C++ rank and a Python rank of the same job always agree on the wire
// format. Conventions (reference: src/stencil.cu:327-464, src/packer.cu:78-82):
//   - a send in direction d happens iff radius(-d) != 0,
//   - the send extent is the RECEIVER's -d halo extent,
//   - cross-rank messages of one (src subdomain, dst subdomain) pair batch
//     into ONE contiguous buffer (16 B-aligned chunks per message x quantity).
#pragma once

#include <algorithm>
#include <map>
#include <set>
#include <tuple>
#include <vector>

#include "stencil_amd/core.hpp"
#include "stencil_amd/placement.hpp"

namespace stencil_amd {

struct PlanMessage {
  Vec3 dir;
  int64_t srcGid, dstGid;
  Vec3 ext; // element extent (equals receiver's halo_extent(-dir))
  int64_t volume() const { return ext.x * ext.y * ext.z; }
};

struct TranslatePlanItem { // same-rank direct-write copy
  int srcLocal, dstLocal;
  Vec3 dir, ext;
};

struct WirePlanItem { // cross-rank packed transfer for one (src, dst) pair
  int peerRank;
  int64_t srcGid, dstGid;
  int localId; // my local domain involved (src for sends, dst for recvs)
  std::vector<PlanMessage> messages;
};

struct ExchangePlan {
  std::vector<TranslatePlanItem> translates;
  std::vector<WirePlanItem> sends, recvs;
};

using HaloExtentFn = Vec3 (*)(const Vec3 &, const Vec3 &, const Radius &);

// deterministic direction sort key (z, y, x lexicographic; planning.py dir_key)
inline bool dir_less(const Vec3 &a, const Vec3 &b) {
  return std::make_tuple(a.z, a.y, a.x) < std::make_tuple(b.z, b.y, b.x);
}

// plan all messages `rank` participates in (mirrors planning.plan_exchange)
inline ExchangePlan plan_exchange(const Placement &placement, const Radius &radius, int rank,
                                  HaloExtentFn haloExtent) {
  ExchangePlan plan;
  const Vec3 dim = placement.dim();
  using Key = std::tuple<int, int64_t, int64_t>; // (peer_rank, src_gid, dst_gid)
  std::map<Key, WirePlanItem> sends, recvs;

  const int nLocal = placement.num_local(rank);
  for (int li = 0; li < nLocal; ++li) {
    const Vec3 myIdx = placement.get_idx(rank, li);
    const int64_t myGid = placement.linearize(myIdx);
    for (int dz = -1; dz <= 1; ++dz)
      for (int dy = -1; dy <= 1; ++dy)
        for (int dx = -1; dx <= 1; ++dx) {
          if (dx == 0 && dy == 0 && dz == 0) continue;
          if (radius.dir(-dx, -dy, -dz) == 0) continue;
          const Vec3 d(dx, dy, dz), neg(-dx, -dy, -dz);

          // --- send to the neighbor at +d ---
          const Vec3 dstIdx((myIdx.x + dx + dim.x) % dim.x, (myIdx.y + dy + dim.y) % dim.y,
                            (myIdx.z + dz + dim.z) % dim.z);
          const int64_t dstGid = placement.linearize(dstIdx);
          const int dstRank = placement.get_rank(dstIdx);
          const Vec3 sExt = haloExtent(neg, placement.subdomain_size(dstIdx), radius);
          if (sExt.x * sExt.y * sExt.z != 0) { // degenerate halos skipped
            if (dstRank == rank) {
              plan.translates.push_back({li, placement.get_subdomain_id(dstIdx), d, sExt});
            } else {
              const Key key{dstRank, myGid, dstGid};
              auto it = sends.find(key);
              if (it == sends.end())
                it = sends.emplace(key, WirePlanItem{dstRank, myGid, dstGid, li, {}}).first;
              it->second.messages.push_back({d, myGid, dstGid, sExt});
            }
          }

          // --- recv from the neighbor at -d (message travels in +d) ---
          const Vec3 srcIdx((myIdx.x - dx + dim.x) % dim.x, (myIdx.y - dy + dim.y) % dim.y,
                            (myIdx.z - dz + dim.z) % dim.z);
          const int64_t srcGid = placement.linearize(srcIdx);
          const int srcRank = placement.get_rank(srcIdx);
          const Vec3 rExt = haloExtent(neg, placement.subdomain_size(myIdx), radius);
          if (srcRank != rank && rExt.x * rExt.y * rExt.z > 0) {
            const Key key{srcRank, srcGid, myGid};
            auto it = recvs.find(key);
            if (it == recvs.end())
              it = recvs.emplace(key, WirePlanItem{srcRank, srcGid, myGid, li, {}}).first;
            it->second.messages.push_back({d, srcGid, myGid, rExt});
          }
        }
  }

  auto emit = [](std::map<Key, WirePlanItem> &items, std::vector<WirePlanItem> &out) {
    for (auto &kv : items) { // std::map iterates keys sorted
      std::stable_sort(kv.second.messages.begin(), kv.second.messages.end(),
                       [](const PlanMessage &a, const PlanMessage &b) { return dir_less(a.dir, b.dir); });
      out.push_back(std::move(kv.second));
    }
  };
  emit(sends, plan.sends);
  emit(recvs, plan.recvs);
  return plan;
}

struct WireChunk {
  int msgIndex;
  int64_t qi, offset, nbytes;
};

// byte layout of one packed buffer (mirrors planning.wire_layout): per
// message (already direction-sorted), per quantity of the exchange group
// (sorted), a 16 B-aligned chunk. Returns total bytes.
inline int64_t wire_layout(const std::vector<PlanMessage> &messages,
                           const std::vector<int64_t> &elemSizes, std::vector<int64_t> qis,
                           std::vector<WireChunk> &chunks) {
  if (qis.empty())
    for (int64_t qi = 0; qi < (int64_t)elemSizes.size(); ++qi) qis.push_back(qi);
  std::sort(qis.begin(), qis.end());
  int64_t off = 0;
  for (size_t mi = 0; mi < messages.size(); ++mi)
    for (int64_t qi : qis) {
      off = (off + 15) / 16 * 16;
      const int64_t nbytes = elemSizes[qi] * messages[mi].volume();
      chunks.push_back({(int)mi, qi, off, nbytes});
      off += nbytes;
    }
  return (off + 15) / 16 * 16;
}

// injective per-rank-pair tags (mirrors planning.pair_seq_tags): index of
// (srcGid, dstGid) in the sorted set of transfers between the two ranks
inline std::map<std::tuple<int, int64_t, int64_t>, int64_t>
pair_seq_tags(const ExchangePlan &plan) {
  std::map<int, std::set<std::pair<int64_t, int64_t>>> byPeer;
  for (const auto *items : {&plan.sends, &plan.recvs})
    for (const WirePlanItem &it : *items) byPeer[it.peerRank].insert({it.srcGid, it.dstGid});
  std::map<std::tuple<int, int64_t, int64_t>, int64_t> seq;
  for (auto &kv : byPeer) {
    int64_t i = 0;
    for (const auto &sd : kv.second) seq[{kv.first, sd.first, sd.second}] = i++;
  }
  return seq;
}

} // namespace stencil_amd
