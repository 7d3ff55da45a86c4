// ExchangeEngine: batched data movement for the halo exchange.
//
// MI355X-native re-design of the reference's transport + packer layer
// (reference: include/stencil/tx_cuda.cuh, packer.cuh, translator.cuh,
// src/packer.cu, src/translator.cu). Instead of one kernel launch per
// message per quantity (reference launches 26 x nQuant kernels per
// exchange), every copy in an exchange is a row in a device-resident job
// table and ONE `copy_batch` kernel per GPU executes the whole table:
//   - "translate" jobs write halo regions directly into the destination
//     domain's memory (same GPU, or a peer GPU over xGMI -- every MI355X
//     pair is one hop, so direct stores replace the reference's
//     pack/cudaMemcpyPeer/unpack staging),
//   - "pack"/"unpack" jobs gather/scatter halo regions into contiguous
//     buffers for RCCL (cross-process) transport.
// Job tables are built once at plan time; they reference the domains'
// device pointer tables (refreshed in place by LocalDomain::swap), so no
// per-iteration rebuild or graph re-capture is needed.
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <vector>

#include <hip/hip_runtime.h>

#include "stencil_amd/core.hpp"
#include "stencil_amd/domain.hpp"

namespace stencil_amd {

// One 3D strided-to-strided copy. Base pointers are resolved at kernel time
// through `slot` indirection (curr/next buffers alternate on swap) or taken
// directly (fixed staging buffers).
struct CopyJob {
  const char *const *srcSlot; // if non-null, base = *srcSlot
  const char *srcDirect;
  char *const *dstSlot;
  char *dstDirect;
  int64_t srcOff, dstOff;     // byte offset of the region start
  int64_t srcPitch, srcPlane; // byte strides between y rows / z planes
  int64_t dstPitch, dstPlane;
  int64_t nWords;  // total words in the region
  int32_t extXw;   // row length in words
  int32_t extY;
  int32_t wordBytes; // 1/2/4/8/16
  int32_t pad_;
};

// A set of jobs launched as one kernel on one device.
struct CopyBatch {
  int dev = -1;
  std::vector<CopyJob> jobs;
  std::vector<int64_t> prefix; // prefix[i] = first block of job i; size nJobs+1
  CopyJob *dJobs = nullptr;
  int64_t *dPrefix = nullptr;
  int64_t nBlocks = 0;
  hipGraphExec_t graphExec = nullptr; // optional hipGraph replay

  void finalize_upload();
  void capture_graph(); // capture the batch launch into a hipGraph
  void launch(hipStream_t stream);
  // plain kernel launch, never via graphExec (safe inside stream capture)
  void launch_plain(hipStream_t stream);
  void destroy();
};

class ExchangeEngine {
public:
  explicit ExchangeEngine(std::vector<std::shared_ptr<LocalDomain>> domains);
  ~ExchangeEngine();

  // enable peer access between every pair of distinct GPUs used
  void enable_peer_all();
  static bool can_access_peer(int src, int dst);

  //// plan-time registration (positions in allocation coordinates)

  // direct-write copy of one region: srcDom's curr -> dstDom's curr
  // (possibly a peer GPU over xGMI). `qis` selects quantities (empty =
  // all); `group` selects the launch group (quantity-group exchanges).
  void add_translate(int srcDom, int dstDom, const Vec3 &srcPos, const Vec3 &dstPos, const Vec3 &ext,
                     int group = 0, const std::vector<int64_t> &qis = {});

  //// cross-process direct writes (HIP IPC over xGMI; the MI355X analog of
  //// the reference's tx_colocated direct-access senders)
  // open a remote rank's domain buffers from their IPC handles; handles are
  // hipIpcMemHandle_t blobs, one per (quantity, parity). `openDev` is the
  // LOCAL device whose kernels will write the remote memory.
  int64_t create_remote_view(int openDev, const std::vector<std::string> &currHandles,
                             const std::vector<std::string> &nextHandles,
                             const std::vector<int64_t> &pitches, const std::vector<int64_t> &ysizes,
                             const std::vector<int64_t> &elemSizes,
                             const std::vector<int64_t> &pads);
  // direct-write one region into a remote view's curr
  void add_translate_view(int srcDom, int64_t view, const Vec3 &srcPos, const Vec3 &dstPos,
                          const Vec3 &ext, int group = 0, const std::vector<int64_t> &qis = {});
  // swap every view's curr/next pointer sets (call in lockstep with
  // LocalDomain::swap on every rank)
  void flip_views();
  // staging buffer on `dom`'s GPU; returns buffer id
  int64_t create_buffer(int dom, int64_t bytes);
  // IPC export/import of staging buffers (for the staged colocated path:
  // thin x-faces pack into the RECEIVER's staging buffer as one contiguous
  // xGMI stream instead of scattered remote stores)
  std::string buffer_ipc_handle(int64_t buf);
  int64_t open_remote_buffer(int openDev, const std::string &handle, int64_t bytes);
  // gather region of quantity qi of dom's curr into buffer at byte offset.
  // `group` selects the launch group: 0 = wire (default), 1/2 = colocated
  // staging parity 0/1 (double-buffered across exchanges).
  void add_pack(int dom, int64_t buf, int64_t offset, const Vec3 &pos, const Vec3 &ext, int64_t qi,
                int group = 0);
  // scatter buffer bytes into region of quantity qi of dom's curr
  void add_unpack(int dom, int64_t buf, int64_t offset, const Vec3 &pos, const Vec3 &ext,
                  int64_t qi, int group = 0);

  // build + upload the per-GPU job tables
  void finalize();

  //// per-exchange execution (stream-ordered; host-sync via the sync_* calls)
  void launch_translates(int group = 0);
  // launch a group's batches onto a caller-owned stream with plain
  // kernel launches (used by whole-step hipGraph capture, where a
  // nested hipGraphLaunch would be illegal)
  void launch_translates_plain_on(uintptr_t stream, int group = 0);
  void launch_packs_plain_on(uintptr_t stream, int group = 0);
  void launch_unpacks_plain_on(uintptr_t stream, int group = 0);
  void launch_packs(int group = 0);
  void launch_unpacks(int group = 0);
  // stream-order every device's group-unpacks after every device's
  // group-packs (events across pack streams) -- needed when a pack on
  // one GPU fills a staging buffer that another GPU's unpack drains
  // (the staged-local cross-device translate path)
  void fence_packs_unpacks(int group = 0);
  // device-side parity flip of every remote view's pointer table (the
  // in-graph analog of flip_views, paired with flip_views_host_only) --
  // swaps devSlots <-> devSlotsAlt contents on `stream`
  void enqueue_view_flips(uintptr_t stream);
  void flip_views_host_only() {
    for (auto &v : views_) v.parity ^= 1;
  }
  void sync_translates();
  void sync_packs(); // also used after unpack
  void sync_all();

  //// compute-stream helpers for apps (streams 0/1 for interior/exterior
  //// overlap; 2/3 exist for concurrent kernel-split experiments)
  hipStream_t compute_stream(int dom, int which = 0);
  uintptr_t compute_stream_handle(int dom) { return (uintptr_t)compute_stream(dom); }
  void sync_compute();
  // make dom's compute stream wait until all exchange streams are idle
  // (used for stream-ordered overlap; v1 exchange is host-synchronous)

  // the per-device pack/unpack stream, exposed so the native RCCL wire
  // can post its grouped send/recv between the pack and unpack batches of
  // the same stream: pack -> wire -> unpack is then fully stream-ordered
  // with a single host sync at the end of the exchange
  uintptr_t pack_stream_handle(int dev) { return (uintptr_t)pack_stream_(dev); }

  //// buffer access (for DLPack export to torch.distributed)
  uintptr_t buffer_ptr(int64_t buf) const { return (uintptr_t)buffers_[buf].ptr; }
  int64_t buffer_bytes(int64_t buf) const { return buffers_[buf].bytes; }
  int buffer_device(int64_t buf) const { return buffers_[buf].dev; }

  LocalDomain &domain(int i) { return *domains_[i]; }
  int num_domains() const { return (int)domains_.size(); }

private:
  struct Buffer {
    char *ptr = nullptr;
    int64_t bytes = 0;
    int dev = -1;
    bool external = false; // IPC-opened (close, don't free)
  };
  struct TranslateSpec {
    int srcDom, dstDom; // dstDom: local domain index, or remote view id
    bool dstIsView = false;
    int group = 0;
    std::vector<int64_t> qis; // empty = all quantities
    Vec3 srcPos, dstPos, ext;
  };
  struct RemoteView {
    int openDev = -1;
    int parity = 0;
    std::vector<char *> base[2]; // [parity][qi] opened pointers
    std::vector<int64_t> pitch, ysize, elemSize, pads;
    char **devSlots = nullptr;    // local device array, refreshed on flip
    char **devSlotsAlt = nullptr; // the other parity (for in-graph flips)
  };
  struct PackSpec {
    int dom;
    int64_t buf, offset;
    Vec3 pos, ext;
    int64_t qi;
    bool unpack;
    int group;
  };

  void build_batches_(const std::vector<TranslateSpec> &ts, const std::vector<PackSpec> &ps);
  hipStream_t comm_stream_(int dev);
  hipStream_t pack_stream_(int dev);

  std::vector<std::shared_ptr<LocalDomain>> domains_;
  std::vector<RemoteView> views_;
  std::vector<Buffer> buffers_;
  std::vector<TranslateSpec> translateSpecs_;
  std::vector<PackSpec> packSpecs_;

  std::map<int, hipStream_t> commStreams_; // per device: translate launches
  std::map<int, hipStream_t> packStreams_; // per device: pack/unpack launches
  std::map<int, hipEvent_t> fenceEvents_;  // per device: pack->unpack fences
  std::vector<hipStream_t> computeStreams_;  // per domain, stream 0
  std::vector<hipStream_t> computeStreams2_; // per domain, stream 1
  std::vector<hipStream_t> computeStreams3_; // streams 2/3: concurrent
  std::vector<hipStream_t> computeStreams4_; // kernel-split experiments

  // pack/unpack launch-group encoding for exchange-group g:
  //   wire = 3g, colo staging parity 0/1 = 3g+1 / 3g+2
  static constexpr int kGroups = 12; // supports 4 exchange groups
  std::vector<CopyBatch> translateBatches_[kGroups];
  std::vector<CopyBatch> packBatches_[kGroups];
  std::vector<CopyBatch> unpackBatches_[kGroups];
  bool finalized_ = false;
};

} // namespace stencil_amd
