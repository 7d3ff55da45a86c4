// RcclWire: torch-free RCCL point-to-point transport for cross-rank halos.
//
// MI355X-native replacement of the reference's MPI data plane
// (reference: src/tx_cuda_aware_mpi.cu:12-76 sends device pointers with
// GPU-aware MPI; tx_cuda.cuh:494-755 stages through pinned host memory).
// On ROCm the equivalent native fabric API is RCCL: one communicator over
// the job, one grouped ncclSend/ncclRecv block per exchange group, posted
// on an engine-owned HIP stream so pack -> wire -> unpack is entirely
// stream-ordered (no host blocking between phases; VERDICT round-1 item 5).
//
// Matching: RCCL pairs the i-th send A->B with the i-th recv on B from A
// (issue order within grouped calls). Ops are therefore sorted by
// (kind, peer, tag) at finalize() with tags computed identically on both
// ranks (planning.pair_seq_tags), which pins the pairing for any number of
// transfers per rank pair.
//
// The unique id travels out-of-band: the Python path broadcasts it over the
// gloo control plane at setup; the pure-C++ path uses a shared file
// (FileBootstrap in distributed.hpp). Neither touches the hot path.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace stencil_amd {

class RcclWire {
public:
  // serialized ncclUniqueId, created once (rank 0) and shared out-of-band
  static std::string unique_id();

  // collective across all `world` ranks (blocks until every rank calls)
  RcclWire(int device, int rank, int world, const std::string &uid);
  ~RcclWire();
  RcclWire(const RcclWire &) = delete;
  RcclWire &operator=(const RcclWire &) = delete;

  // plan-time registration; tag orders transfers within a rank pair
  void add_send(int group, uintptr_t ptr, int64_t bytes, int peer, int64_t tag);
  void add_recv(int group, uintptr_t ptr, int64_t bytes, int peer, int64_t tag);
  void finalize();

  // enqueue one exchange group's sends+recvs on `stream` (grouped);
  // completion is stream-ordered, no host sync inside
  void post(int group, uintptr_t stream);

  // 1-element all-reduce on a preallocated device scratch word: a device
  // barrier over this communicator (used for the colocated-rank barrier
  // when every rank drives a distinct GPU)
  void barrier(uintptr_t stream);

  int rank() const { return rank_; }
  int world() const { return world_; }
  int device() const { return dev_; }

private:
  struct Op {
    bool send;
    uintptr_t ptr;
    int64_t bytes;
    int peer;
    int64_t tag;
  };
  void *comm_ = nullptr; // ncclComm_t
  float *scratch_ = nullptr;
  int dev_, rank_, world_;
  std::vector<std::vector<Op>> ops_;
  bool finalized_ = false;
};

} // namespace stencil_amd
