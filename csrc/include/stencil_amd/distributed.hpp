// DistributedDomain, C++ side: the full library (partition, placement,
// planning, transport) without Python.
//
// MI355X-native re-design of the reference's C++ app-facing API
// (reference: include/stencil/stencil.hpp:114-217, src/stencil.cu). The
// Python DistributedDomain (stencil_amd/core.py) drives the same
// LocalDomain/ExchangeEngine/RcclWire pieces through the same planner
// (planning.hpp mirrors planning.py exactly), so a C++ app gets identical
// partitioning, placement and wire formats.
//
// Process model:
//   - world = 1 (default): one process, any number of GPUs (set_gpus);
//     every halo moves by direct-write translate kernels over xGMI.
//   - world > 1: one process per GPU. Rank/world come from STENCIL_RANK /
//     STENCIL_WORLD (or RANK/WORLD_SIZE), and the control plane (slot
//     gather + RCCL unique id) is a FileBootstrap over the shared
//     directory STENCIL_BOOTSTRAP_DIR — no MPI, no torchrun. Cross-rank
//     halos are packed and moved by RcclWire (grouped ncclSend/ncclRecv
//     over xGMI), stream-ordered pack -> wire -> unpack on the engine's
//     pack stream. (The Python orchestrator additionally offers the
//     colocated HIP-IPC direct-write path; the C++ wire path is the
//     RCCL one.)
#pragma once

#include <map>
#include <memory>
#include <string>
#include <vector>

#include "stencil_amd/core.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/placement.hpp"
#include "stencil_amd/planning.hpp"
#include "stencil_amd/wire.hpp"

namespace stencil_amd {

// allgather of byte-strings via a shared directory (atomic rename +
// poll): the out-of-band control plane for multi-process C++ runs.
// Python ranks use torch.distributed(gloo) instead; both are setup-only.
class FileBootstrap {
public:
  FileBootstrap(std::string dir, int rank, int world);
  // every rank calls with its payload; returns all `world` payloads.
  // `phase` must be unique per collective (and identical across ranks).
  std::vector<std::string> allgather(const std::string &phase, const std::string &payload);

private:
  std::string dir_;
  int rank_, world_;
  int seq_ = 0; // per-call suffix: repeated collectives never collide
};

class DistributedDomain {
public:
  DistributedDomain(int64_t x, int64_t y, int64_t z);

  //// configuration (before realize)
  void set_radius(const Radius &r) { radius_ = r; }
  void set_radius(int64_t r) { radius_ = Radius::constant(r); }
  int64_t add_data(int64_t elemSize, const std::string &name = "");
  template <typename T> int64_t add_data(const std::string &name = "") {
    return add_data((int64_t)sizeof(T), name);
  }
  void set_gpus(const std::vector<int> &gpus) { gpus_ = gpus; }
  void set_placement(PlacementStrategy s) { strategy_ = s; }
  // partition quantities into independently exchangeable groups
  void set_exchange_groups(const std::vector<std::vector<int64_t>> &groups);

  void realize();

  //// per-iteration
  void exchange(int group = 0);
  void exchange_begin(int group = 0); // translates + packs enqueued, returns
  void exchange_end(int group = 0);   // wire + unpack + one host sync
  void swap();

  //// queries
  int rank() const { return rank_; }
  int world() const { return world_; }
  int num_local() const { return (int)domains_.size(); }
  Vec3 size() const { return size_; }
  Rect3 compute_region() const { return Rect3(Vec3(0, 0, 0), size_); }
  // global-coordinate box of local subdomain li
  Rect3 local_rect(int li) const;
  // per local domain: the sub-box whose stencil never reads halo
  std::vector<Rect3> get_interior() const;
  // per local domain: non-overlapping slabs covering local_rect minus interior
  std::vector<std::vector<Rect3>> get_exterior() const;

  LocalDomain &domain(int li) { return *domains_[li]; }
  ExchangeEngine &engine() { return *engine_; }
  const Placement &placement() const { return *placement_; }

  // exchanged bytes per transport per full exchange (all groups)
  int64_t bytes_translate() const { return bytesTranslate_; }
  int64_t bytes_wire() const { return bytesWire_; }

  // dump each local subdomain interior as CSV 'Z,Y,X,q0,q1,...'
  // (reference src/stencil.cu:1188-1264; same format as the Python
  // write_paraview)
  void write_paraview(const std::string &prefix);

  // setup-phase seconds (reference STENCIL_SETUP_STATS,
  // stencil.hpp:103-112): keys topo/placement/realize/plan/create
  const std::map<std::string, double> &setup_times() const { return setupTimes_; }

private:
  void gather_slots_(std::vector<Slot> &slots);

  Vec3 size_;
  Radius radius_ = Radius::constant(0);
  std::vector<std::pair<int64_t, std::string>> data_;
  std::vector<std::vector<int64_t>> groups_;
  std::vector<int> gpus_;
  PlacementStrategy strategy_ = PlacementStrategy::NodeAware;

  int rank_ = 0, world_ = 1;
  std::unique_ptr<FileBootstrap> boot_;
  std::unique_ptr<Placement> placement_;
  std::vector<std::shared_ptr<LocalDomain>> domains_;
  std::unique_ptr<ExchangeEngine> engine_;
  std::unique_ptr<RcclWire> wire_;
  int wireDev_ = -1;
  std::vector<bool> hasWire_;
  int64_t bytesTranslate_ = 0, bytesWire_ = 0;
  std::map<std::string, double> setupTimes_;
  bool realized_ = false;
};

} // namespace stencil_amd
