// RcclWire implementation (see wire.hpp for the design rationale).
#include "stencil_amd/wire.hpp"
#include "stencil_amd/hip_check.hpp"

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include <algorithm>
#include <cstring>
#include <stdexcept>

namespace stencil_amd {

namespace {
void nccl_check(ncclResult_t r, const char *what) {
  if (r != ncclSuccess)
    throw std::runtime_error(std::string("RCCL error in ") + what + ": " +
                             ncclGetErrorString(r));
}
#define STENCIL_NCCL(call) nccl_check((call), #call)
constexpr int kMaxGroups = 4;
} // namespace

std::string RcclWire::unique_id() {
  ncclUniqueId id;
  STENCIL_NCCL(ncclGetUniqueId(&id));
  return std::string((const char *)&id, sizeof(id));
}

RcclWire::RcclWire(int device, int rank, int world, const std::string &uid)
    : dev_(device), rank_(rank), world_(world), ops_(kMaxGroups) {
  if (uid.size() != sizeof(ncclUniqueId))
    throw std::runtime_error("RcclWire: bad unique id size");
  ncclUniqueId id;
  std::memcpy(&id, uid.data(), sizeof(id));
  STENCIL_HIP(hipSetDevice(dev_));
  ncclComm_t comm = nullptr;
  STENCIL_NCCL(ncclCommInitRank(&comm, world_, id, rank_));
  comm_ = comm;
  STENCIL_HIP(hipMalloc((void **)&scratch_, sizeof(float)));
  STENCIL_HIP(hipMemset(scratch_, 0, sizeof(float)));
}

RcclWire::~RcclWire() {
  if (comm_) (void)ncclCommDestroy((ncclComm_t)comm_);
  if (scratch_) {
    (void)hipSetDevice(dev_);
    (void)hipFree(scratch_);
  }
}

void RcclWire::add_send(int group, uintptr_t ptr, int64_t bytes, int peer, int64_t tag) {
  if (finalized_) throw std::runtime_error("RcclWire: add after finalize");
  ops_.at(group).push_back({true, ptr, bytes, peer, tag});
}

void RcclWire::add_recv(int group, uintptr_t ptr, int64_t bytes, int peer, int64_t tag) {
  if (finalized_) throw std::runtime_error("RcclWire: add after finalize");
  ops_.at(group).push_back({false, ptr, bytes, peer, tag});
}

void RcclWire::finalize() {
  // deterministic issue order: all sends (by peer, tag), then all recvs
  // (by peer, tag). Both ranks of a pair compute identical tags, so the
  // i-th send A->B lines up with the i-th recv on B from A.
  for (auto &g : ops_)
    std::stable_sort(g.begin(), g.end(), [](const Op &a, const Op &b) {
      if (a.send != b.send) return a.send > b.send;
      if (a.peer != b.peer) return a.peer < b.peer;
      return a.tag < b.tag;
    });
  finalized_ = true;
}

void RcclWire::post(int group, uintptr_t stream) {
  const auto &g = ops_.at(group);
  if (g.empty()) return;
  STENCIL_HIP(hipSetDevice(dev_));
  STENCIL_NCCL(ncclGroupStart());
  for (const Op &op : g) {
    if (op.send)
      STENCIL_NCCL(ncclSend((const void *)op.ptr, (size_t)op.bytes, ncclChar, op.peer,
                            (ncclComm_t)comm_, (hipStream_t)stream));
    else
      STENCIL_NCCL(ncclRecv((void *)op.ptr, (size_t)op.bytes, ncclChar, op.peer,
                            (ncclComm_t)comm_, (hipStream_t)stream));
  }
  STENCIL_NCCL(ncclGroupEnd());
}

void RcclWire::barrier(uintptr_t stream) {
  STENCIL_HIP(hipSetDevice(dev_));
  STENCIL_NCCL(ncclAllReduce(scratch_, scratch_, 1, ncclFloat, ncclSum, (ncclComm_t)comm_,
                             (hipStream_t)stream));
}

} // namespace stencil_amd
