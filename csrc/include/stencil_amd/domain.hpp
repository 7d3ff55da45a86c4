// LocalDomain: one GPU's subdomain of the distributed grid.
//
// MI355X-native re-design of the reference's LocalDomain
// (reference: include/stencil/local_domain.cuh, src/local_domain.cu).
// Double-buffered (curr/next) per-quantity allocations with rows padded to a
// 256 B pitch for coalesced HBM3E access, plus device-resident arrays of the
// raw base pointers. Unlike the reference (which swaps the device-array
// POINTERS on swap()), swap() here refreshes the CONTENTS of two fixed
// device arrays, so pre-built copy-job tables and captured hipGraphs remain
// valid across iterations.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

#include <hip/hip_runtime.h>

#include "stencil_amd/core.hpp"

namespace stencil_amd {

// a pitched 3D allocation: rows of `pitch` bytes, `ysize` rows per z-plane
struct Pitched {
  char *ptr = nullptr;
  int64_t pitch = 0; // bytes between consecutive y rows
  int64_t ysize = 0; // rows per z-plane
  int64_t plane() const { return pitch * ysize; }
};

class LocalDomain {
public:
  LocalDomain(const Vec3 &sz, const Vec3 &origin, int dev);
  ~LocalDomain();
  LocalDomain(const LocalDomain &) = delete;
  LocalDomain &operator=(const LocalDomain &) = delete;

  int64_t add_data(int64_t elemSize, const std::string &name = "");
  void set_radius(const Radius &r) { radius_ = r; }
  const Radius &radius() const { return radius_; }

  // allocate device memory (hipMalloc) and upload pointer tables
  void realize();
  // swap curr/next and refresh the device pointer tables in place
  void swap();
  // swap only the HOST-side mirrors: used by the whole-step hipGraph
  // path, whose in-graph swap_tables kernel flips the device tables
  void swap_host_only() { std::swap(curr_, next_); }
  // enqueue a tiny kernel that flips the CONTENTS of the device pointer
  // tables (curr <-> next) on `stream` -- the graph-node counterpart of
  // swap(): slot-indirect copy jobs stay valid across replays
  void enqueue_table_swap(hipStream_t stream);

  //// geometry (all positions in "allocation coordinates": element offsets
  //// from the first allocated element, which sits at global coordinate
  //// origin - (radius.x(-1), radius.y(-1), radius.z(-1)))

  // position of the halo (halo=true) or adjacent-interior (halo=false)
  // region on side `dir`; dir=0 gives the whole compute region.
  // Convention matches the reference (src/local_domain.cu:86-129): a message
  // sent in direction d packs halo_pos(d, false) with extent
  // halo_extent(-d) and lands in the receiver's halo_pos(-d, true).
  static Vec3 halo_pos(const Vec3 &dir, const Vec3 &sz, const Radius &radius, bool halo);
  Vec3 halo_pos(const Vec3 &dir, bool halo) const { return halo_pos(dir, sz_, radius_, halo); }

  static Vec3 halo_extent(const Vec3 &dir, const Vec3 &sz, const Radius &radius) {
    Vec3 ret;
    ret.x = (0 == dir.x) ? sz.x : radius.x((int)dir.x);
    ret.y = (0 == dir.y) ? sz.y : radius.y((int)dir.y);
    ret.z = (0 == dir.z) ? sz.z : radius.z((int)dir.z);
    return ret;
  }
  Vec3 halo_extent(const Vec3 &dir) const { return halo_extent(dir, sz_, radius_); }

  // halo region on side `dir` in GLOBAL coordinates
  Rect3 halo_coords(const Vec3 &dir, bool halo) const;

  Rect3 compute_region() const { return Rect3(origin_, origin_ + sz_); }
  Rect3 full_region() const;

  Vec3 raw_size() const {
    return Vec3(sz_.x + radius_.x(-1) + radius_.x(1), sz_.y + radius_.y(-1) + radius_.y(1),
                sz_.z + radius_.z(-1) + radius_.z(1));
  }

  int64_t halo_bytes(const Vec3 &dir, int64_t qi) const {
    return elemSize_.at(qi) * halo_extent(dir).flatten();
  }

  //// accessors
  const Vec3 &size() const { return sz_; }
  const Vec3 &origin() const { return origin_; }
  int gpu() const { return dev_; }
  int64_t num_data() const { return (int64_t)elemSize_.size(); }
  int64_t elem_size(int64_t qi) const { return elemSize_[qi]; }
  const std::string &name(int64_t qi) const { return name_[qi]; }
  const Pitched &curr(int64_t qi) const { return curr_[qi]; }
  const Pitched &next(int64_t qi) const { return next_[qi]; }
  // device arrays (nq entries) holding the current/next raw base pointers;
  // fixed addresses for the lifetime of the domain
  char **dev_curr_slots() const { return devCurrRaw_; }
  char **dev_next_slots() const { return devNextRaw_; }

  // IPC export of a quantity buffer (hipIpcMemHandle_t blob) for the
  // cross-process direct-write transport. The handle refers to the
  // allocation BASE; the importer must add pad_bytes(qi) to reach
  // element (0,0,0).
  std::string ipc_handle(int64_t qi, bool next) const;
  // leading alignment pad (so the interior x-start is 16 B aligned and
  // vectorized stencil kernels have no head/tail cells)
  int64_t pad_bytes(int64_t qi) const { return padBytes_.at(qi); }

  // blocking element-region copies (pos in allocation coords)
  void region_to_host(void *dst, const Vec3 &pos, const Vec3 &ext, int64_t qi, bool fromNext = false) const;
  void region_from_host(const void *src, const Vec3 &pos, const Vec3 &ext, int64_t qi, bool toNext = false) const;

private:
  void swapUpload_(); // refresh device pointer tables from host vectors

  Vec3 sz_;
  Vec3 origin_;
  Radius radius_ = Radius::constant(0);
  int dev_;
  bool realized_ = false;

  std::vector<int64_t> elemSize_;
  std::vector<std::string> name_;
  std::vector<Pitched> curr_, next_;
  std::vector<int64_t> padBytes_;      // leading pad per quantity
  std::vector<char *> allocBases_;     // hipMalloc bases (curr then next)
  char **devCurrRaw_ = nullptr; // device array of curr base ptrs
  char **devNextRaw_ = nullptr;
};

} // namespace stencil_amd
