// ExchangeEngine implementation: job-table construction and the batched
// copy kernel. See engine.hpp for the design rationale.
#include "stencil_amd/device_util.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"

#include <roctracer/roctx.h>

#include <algorithm>
#include <cstdlib>
#include <cstring>
#include <initializer_list>
#include <numeric>
#include <stdexcept>

namespace stencil_amd {

namespace {

constexpr int kBlock = 256;

struct alignas(16) W16 {
  uint64_t a, b;
};

__global__ void copy_batch_kernel(const CopyJob *__restrict__ jobs, const int64_t *__restrict__ prefix,
                                  int nJobs) {
  // binary-search the job that owns this block
  const int64_t b = blockIdx.x;
  int lo = 0, hi = nJobs - 1;
  while (lo < hi) {
    const int mid = (lo + hi) / 2;
    if (prefix[mid + 1] <= b) {
      lo = mid + 1;
    } else {
      hi = mid;
    }
  }
  const CopyJob j = jobs[lo];
  const char *src = uniform_ptr((j.srcSlot ? *j.srcSlot : j.srcDirect) + j.srcOff);
  char *dst = uniform_ptr((j.dstSlot ? *j.dstSlot : j.dstDirect) + j.dstOff);

  const int64_t jobBlocks = prefix[lo + 1] - prefix[lo];
  const int64_t stride = jobBlocks * blockDim.x;
  for (int64_t w = (b - prefix[lo]) * blockDim.x + threadIdx.x; w < j.nWords; w += stride) {
    const int32_t wx = (int32_t)(w % j.extXw);
    const int64_t t = w / j.extXw;
    const int32_t wy = (int32_t)(t % j.extY);
    const int64_t wz = t / j.extY;
    const char *s = src + wz * j.srcPlane + (int64_t)wy * j.srcPitch + (int64_t)wx * j.wordBytes;
    char *d = dst + wz * j.dstPlane + (int64_t)wy * j.dstPitch + (int64_t)wx * j.wordBytes;
    switch (j.wordBytes) {
    case 16:
      *reinterpret_cast<W16 *>(d) = *reinterpret_cast<const W16 *>(s);
      break;
    case 8:
      *reinterpret_cast<uint64_t *>(d) = *reinterpret_cast<const uint64_t *>(s);
      break;
    case 4:
      *reinterpret_cast<uint32_t *>(d) = *reinterpret_cast<const uint32_t *>(s);
      break;
    case 2:
      *reinterpret_cast<uint16_t *>(d) = *reinterpret_cast<const uint16_t *>(s);
      break;
    default:
      *d = *s;
      break;
    }
  }
}

// largest word size (<=16) honoring every alignment constraint of the copy
int pick_word(int64_t rowBytes, std::initializer_list<int64_t> alignedQuantities) {
  for (int w : {16, 8, 4, 2}) {
    if (rowBytes % w) continue;
    bool ok = true;
    for (int64_t q : alignedQuantities)
      if (q % w) {
        ok = false;
        break;
      }
    if (ok) return w;
  }
  return 1;
}

} // namespace

void CopyBatch::finalize_upload() {
  // Cap the batch at ~512 blocks (grid-stride covers the rest): a halo
  // exchange overlapped with compute need not flood all 256 CUs. One
  // measurement showed +8.6% on the jacobi step, a re-measurement on
  // another box showed no difference (run-to-run DVFS variance ~8%); the
  // cap is kept as it is never worse and bounds CU contention by
  // construction. STENCIL_AMD_COPY_DIV overrides for experiments.
  constexpr int64_t kMaxBlocks = 512;
  int64_t div = 0;
  if (const char *e = getenv("STENCIL_AMD_COPY_DIV")) div = atoll(e);
  if (div < 1) {
    int64_t natural = 0;
    for (const auto &j : jobs) natural += std::max<int64_t>(1, (j.nWords + kBlock - 1) / kBlock);
    div = std::max<int64_t>(1, (natural + kMaxBlocks - 1) / kMaxBlocks);
  }
  prefix.assign(jobs.size() + 1, 0);
  for (size_t i = 0; i < jobs.size(); ++i) {
    const int64_t blocks =
        std::max<int64_t>(1, (jobs[i].nWords + kBlock * div - 1) / (kBlock * div));
    prefix[i + 1] = prefix[i] + blocks;
  }
  nBlocks = prefix.back();
  STENCIL_HIP(hipSetDevice(dev));
  STENCIL_HIP(hipMalloc((void **)&dJobs, jobs.size() * sizeof(CopyJob)));
  STENCIL_HIP(hipMalloc((void **)&dPrefix, prefix.size() * sizeof(int64_t)));
  STENCIL_HIP(hipMemcpy(dJobs, jobs.data(), jobs.size() * sizeof(CopyJob), hipMemcpyHostToDevice));
  STENCIL_HIP(
      hipMemcpy(dPrefix, prefix.data(), prefix.size() * sizeof(int64_t), hipMemcpyHostToDevice));
}

// Optional: wrap the batch launch in a hipGraph and replay it. With the
// job-table design an exchange is already 1-2 launches per GPU (the
// reference needed graphs to amortize its 26 x nQuant launch storm,
// src/packer.cu:96-106), so this mainly shaves host launch latency.
// Enabled by STENCIL_AMD_GRAPHS=1.
void CopyBatch::capture_graph() {
  if (jobs.empty()) return;
  STENCIL_HIP(hipSetDevice(dev));
  hipStream_t cap;
  STENCIL_HIP(hipStreamCreateWithFlags(&cap, hipStreamNonBlocking));
  STENCIL_HIP(hipStreamBeginCapture(cap, hipStreamCaptureModeThreadLocal));
  hipLaunchKernelGGL(copy_batch_kernel, dim3((uint32_t)nBlocks), dim3(kBlock), 0, cap, dJobs,
                     dPrefix, (int)jobs.size());
  hipGraph_t graph;
  STENCIL_HIP(hipStreamEndCapture(cap, &graph));
  STENCIL_HIP(hipGraphInstantiate(&graphExec, graph, nullptr, nullptr, 0));
  STENCIL_HIP(hipGraphDestroy(graph));
  STENCIL_HIP(hipStreamDestroy(cap));
}

void CopyBatch::launch(hipStream_t stream) {
  if (jobs.empty()) return;
  STENCIL_HIP(hipSetDevice(dev));
  if (graphExec) {
    STENCIL_HIP(hipGraphLaunch(graphExec, stream));
    return;
  }
  hipLaunchKernelGGL(copy_batch_kernel, dim3((uint32_t)nBlocks), dim3(kBlock), 0, stream, dJobs,
                     dPrefix, (int)jobs.size());
  STENCIL_HIP(hipGetLastError());
}

void CopyBatch::launch_plain(hipStream_t stream) {
  if (jobs.empty()) return;
  hipLaunchKernelGGL(copy_batch_kernel, dim3((uint32_t)nBlocks), dim3(kBlock), 0, stream, dJobs,
                     dPrefix, (int)jobs.size());
  STENCIL_HIP(hipGetLastError());
}

void CopyBatch::destroy() {
  if (graphExec) (void)hipGraphExecDestroy(graphExec);
  graphExec = nullptr;
  if (dJobs) (void)hipFree(dJobs);
  if (dPrefix) (void)hipFree(dPrefix);
  dJobs = nullptr;
  dPrefix = nullptr;
}

ExchangeEngine::ExchangeEngine(std::vector<std::shared_ptr<LocalDomain>> domains)
    : domains_(std::move(domains)) {
  computeStreams_.resize(domains_.size(), nullptr);
  computeStreams2_.resize(domains_.size(), nullptr);
  computeStreams3_.resize(domains_.size(), nullptr);
  computeStreams4_.resize(domains_.size(), nullptr);
  // spin-wait host syncs: interrupt-based stream-sync wakeups cost
  // 20-100 us each on ROCm; an HPC bench prefers burning the core.
  // Tolerant: the flag may be rejected once a device context exists.
  const char *spin = getenv("STENCIL_AMD_SPIN");
  if (!spin || spin[0] != '0') {
    for (auto &d : domains_) {
      if (hipSetDevice(d->gpu()) != hipSuccess) continue;
      (void)hipSetDeviceFlags(hipDeviceScheduleSpin);
      (void)hipGetLastError(); // clear hipErrorSetOnActiveProcess
    }
  }
}

ExchangeEngine::~ExchangeEngine() {
  for (int g = 0; g < kGroups; ++g) {
    for (auto &b : translateBatches_[g]) b.destroy();
    for (auto &b : packBatches_[g]) b.destroy();
    for (auto &b : unpackBatches_[g]) b.destroy();
  }
  for (auto &kv : commStreams_) (void)hipStreamDestroy(kv.second);
  for (auto &kv : packStreams_) (void)hipStreamDestroy(kv.second);
  for (auto &kv : fenceEvents_) (void)hipEventDestroy(kv.second);
  for (auto *vec : {&computeStreams_, &computeStreams2_, &computeStreams3_, &computeStreams4_})
    for (auto s : *vec)
      if (s) (void)hipStreamDestroy(s);
  for (auto &b : buffers_) {
    if (!b.ptr) continue;
    if (b.external) {
      (void)hipSetDevice(b.dev);
      (void)hipIpcCloseMemHandle(b.ptr);
    } else {
      (void)hipFree(b.ptr);
    }
  }
  for (auto &v : views_) {
    (void)hipSetDevice(v.openDev);
    for (int par = 0; par < 2; ++par)
      for (size_t qi = 0; qi < v.base[par].size(); ++qi)
        if (v.base[par][qi]) (void)hipIpcCloseMemHandle(v.base[par][qi] - v.pads[qi]);
    if (v.devSlots) (void)hipFree(v.devSlots);
    if (v.devSlotsAlt) (void)hipFree(v.devSlotsAlt);
  }
}

bool ExchangeEngine::can_access_peer(int src, int dst) {
  if (src == dst) return true;
  int ok = 0;
  if (hipDeviceCanAccessPeer(&ok, src, dst) != hipSuccess) return false;
  return ok != 0;
}

void ExchangeEngine::enable_peer_all() {
  std::vector<int> devs;
  for (auto &d : domains_) devs.push_back(d->gpu());
  std::sort(devs.begin(), devs.end());
  devs.erase(std::unique(devs.begin(), devs.end()), devs.end());
  for (int a : devs)
    for (int b : devs) {
      if (a == b) continue;
      STENCIL_HIP(hipSetDevice(a));
      hipError_t err = hipDeviceEnablePeerAccess(b, 0);
      if (err != hipSuccess && err != hipErrorPeerAccessAlreadyEnabled) {
        STENCIL_HIP(err);
      }
      (void)hipGetLastError(); // clear sticky already-enabled
    }
}

void ExchangeEngine::add_translate(int srcDom, int dstDom, const Vec3 &srcPos, const Vec3 &dstPos,
                                   const Vec3 &ext, int group, const std::vector<int64_t> &qis) {
  translateSpecs_.push_back({srcDom, dstDom, false, group, qis, srcPos, dstPos, ext});
}

int64_t ExchangeEngine::create_remote_view(int openDev, const std::vector<std::string> &currHandles,
                                           const std::vector<std::string> &nextHandles,
                                           const std::vector<int64_t> &pitches,
                                           const std::vector<int64_t> &ysizes,
                                           const std::vector<int64_t> &elemSizes,
                                           const std::vector<int64_t> &pads) {
  const size_t nq = currHandles.size();
  if (nextHandles.size() != nq || pitches.size() != nq || ysizes.size() != nq ||
      elemSizes.size() != nq || pads.size() != nq)
    throw std::runtime_error("create_remote_view: size mismatch");
  RemoteView v;
  v.openDev = openDev;
  v.pitch = pitches;
  v.ysize = ysizes;
  v.elemSize = elemSizes;
  v.pads = pads;
  STENCIL_HIP(hipSetDevice(openDev));
  auto open_one = [&](const std::string &blob) {
    if (blob.size() != sizeof(hipIpcMemHandle_t))
      throw std::runtime_error("create_remote_view: bad handle size");
    hipIpcMemHandle_t h;
    std::memcpy(&h, blob.data(), sizeof(h));
    void *p = nullptr;
    STENCIL_HIP(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess));
    return (char *)p;
  };
  for (size_t qi = 0; qi < nq; ++qi) {
    // the IPC handle maps the allocation base; element (0,0,0) sits
    // pads[qi] bytes in (LocalDomain alignment pad)
    v.base[0].push_back(open_one(currHandles[qi]) + pads[qi]);
    v.base[1].push_back(open_one(nextHandles[qi]) + pads[qi]);
  }
  STENCIL_HIP(hipMalloc((void **)&v.devSlots, nq * sizeof(char *)));
  STENCIL_HIP(
      hipMemcpy(v.devSlots, v.base[0].data(), nq * sizeof(char *), hipMemcpyHostToDevice));
  STENCIL_HIP(hipMalloc((void **)&v.devSlotsAlt, nq * sizeof(char *)));
  STENCIL_HIP(
      hipMemcpy(v.devSlotsAlt, v.base[1].data(), nq * sizeof(char *), hipMemcpyHostToDevice));
  views_.push_back(std::move(v));
  return (int64_t)views_.size() - 1;
}

void ExchangeEngine::add_translate_view(int srcDom, int64_t view, const Vec3 &srcPos,
                                        const Vec3 &dstPos, const Vec3 &ext, int group,
                                        const std::vector<int64_t> &qis) {
  translateSpecs_.push_back({srcDom, (int)view, true, group, qis, srcPos, dstPos, ext});
}

void ExchangeEngine::flip_views() {
  for (auto &v : views_) {
    v.parity ^= 1;
    STENCIL_HIP(hipSetDevice(v.openDev));
    STENCIL_HIP(hipMemcpy(v.devSlots, v.base[v.parity].data(),
                          v.base[v.parity].size() * sizeof(char *), hipMemcpyHostToDevice));
    STENCIL_HIP(hipMemcpy(v.devSlotsAlt, v.base[v.parity ^ 1].data(),
                          v.base[v.parity ^ 1].size() * sizeof(char *), hipMemcpyHostToDevice));
  }
}

namespace {
__global__ void swap_slots_kernel(char **a, char **b, int n) {
  const int i = threadIdx.x;
  if (i < n) {
    char *t = a[i];
    a[i] = b[i];
    b[i] = t;
  }
}
} // namespace

void ExchangeEngine::enqueue_view_flips(uintptr_t stream) {
  for (auto &v : views_) {
    hipLaunchKernelGGL(swap_slots_kernel, dim3(1), dim3(256), 0, (hipStream_t)stream, v.devSlots,
                       v.devSlotsAlt, (int)v.base[0].size());
    STENCIL_HIP(hipGetLastError());
  }
}

int64_t ExchangeEngine::create_buffer(int dom, int64_t bytes) {
  Buffer b;
  b.dev = domains_[dom]->gpu();
  b.bytes = bytes;
  STENCIL_HIP(hipSetDevice(b.dev));
  STENCIL_HIP(hipMalloc((void **)&b.ptr, std::max<int64_t>(bytes, 16)));
  buffers_.push_back(b);
  return (int64_t)buffers_.size() - 1;
}

void ExchangeEngine::add_pack(int dom, int64_t buf, int64_t offset, const Vec3 &pos, const Vec3 &ext,
                              int64_t qi, int group) {
  packSpecs_.push_back({dom, buf, offset, pos, ext, qi, false, group});
}

void ExchangeEngine::add_unpack(int dom, int64_t buf, int64_t offset, const Vec3 &pos,
                                const Vec3 &ext, int64_t qi, int group) {
  packSpecs_.push_back({dom, buf, offset, pos, ext, qi, true, group});
}

std::string ExchangeEngine::buffer_ipc_handle(int64_t buf) {
  STENCIL_HIP(hipSetDevice(buffers_[buf].dev));
  hipIpcMemHandle_t h;
  STENCIL_HIP(hipIpcGetMemHandle(&h, buffers_[buf].ptr));
  return std::string((const char *)&h, sizeof(h));
}

int64_t ExchangeEngine::open_remote_buffer(int openDev, const std::string &handle, int64_t bytes) {
  if (handle.size() != sizeof(hipIpcMemHandle_t))
    throw std::runtime_error("open_remote_buffer: bad handle size");
  STENCIL_HIP(hipSetDevice(openDev));
  hipIpcMemHandle_t h;
  std::memcpy(&h, handle.data(), sizeof(h));
  void *p = nullptr;
  STENCIL_HIP(hipIpcOpenMemHandle(&p, h, hipIpcMemLazyEnablePeerAccess));
  Buffer b;
  b.ptr = (char *)p;
  b.bytes = bytes;
  b.dev = openDev;
  b.external = true; // close, don't free
  buffers_.push_back(b);
  return (int64_t)buffers_.size() - 1;
}

void ExchangeEngine::finalize() {
  if (finalized_) throw std::runtime_error("ExchangeEngine::finalize called twice");
  build_batches_(translateSpecs_, packSpecs_);
  const char *g = getenv("STENCIL_AMD_GRAPHS");
  const bool graphs = g && g[0] == '1';
  std::vector<std::vector<CopyBatch> *> sets;
  for (int gr = 0; gr < kGroups; ++gr) {
    sets.push_back(&translateBatches_[gr]);
    sets.push_back(&packBatches_[gr]);
    sets.push_back(&unpackBatches_[gr]);
  }
  for (auto *set : sets)
    for (auto &b : *set) {
      b.finalize_upload();
      if (graphs) b.capture_graph();
    }
  finalized_ = true;
}

void ExchangeEngine::build_batches_(const std::vector<TranslateSpec> &ts,
                                    const std::vector<PackSpec> &ps) {
  std::map<std::pair<int, int>, CopyBatch> tb, pb, ub; // (group, gpu)

  for (const auto &t : ts) {
    LocalDomain &s = *domains_[t.srcDom];
    std::vector<int64_t> qis = t.qis;
    if (qis.empty())
      for (int64_t qi = 0; qi < s.num_data(); ++qi) qis.push_back(qi);
    for (int64_t qi : qis) {
      const int64_t es = s.elem_size(qi);
      const Pitched &sp = s.curr(qi); // pitch/ysize only; base via slot
      int64_t dPitch, dPlane;
      char *const *dSlot;
      if (t.dstIsView) {
        RemoteView &v = views_[t.dstDom];
        if (v.elemSize[qi] != es) throw std::runtime_error("view elem size mismatch");
        dPitch = v.pitch[qi];
        dPlane = v.pitch[qi] * v.ysize[qi];
        dSlot = (char *const *)(v.devSlots + qi);
      } else {
        LocalDomain &d = *domains_[t.dstDom];
        const Pitched &dp = d.curr(qi);
        dPitch = dp.pitch;
        dPlane = dp.plane();
        dSlot = (char *const *)(d.dev_curr_slots() + qi);
      }
      CopyJob j{};
      j.srcSlot = (const char *const *)(s.dev_curr_slots() + qi);
      j.dstSlot = dSlot;
      j.srcOff = t.srcPos.z * sp.plane() + t.srcPos.y * sp.pitch + t.srcPos.x * es;
      j.dstOff = t.dstPos.z * dPlane + t.dstPos.y * dPitch + t.dstPos.x * es;
      j.srcPitch = sp.pitch;
      j.srcPlane = sp.plane();
      j.dstPitch = dPitch;
      j.dstPlane = dPlane;
      const int64_t rowBytes = t.ext.x * es;
      const int w = pick_word(rowBytes, {j.srcOff, j.dstOff, j.srcPitch, j.dstPitch});
      j.wordBytes = w;
      j.extXw = (int32_t)(rowBytes / w);
      j.extY = (int32_t)t.ext.y;
      j.nWords = (int64_t)j.extXw * t.ext.y * t.ext.z;
      auto &batch = tb[{t.group, s.gpu()}];
      batch.dev = s.gpu();
      batch.jobs.push_back(j);
    }
  }

  for (const auto &p : ps) {
    LocalDomain &dom = *domains_[p.dom];
    const int64_t es = dom.elem_size(p.qi);
    const Pitched &dp = dom.curr(p.qi);
    const Buffer &buf = buffers_[p.buf];
    const int64_t rowBytes = p.ext.x * es;
    CopyJob j{};
    const int64_t domOff = p.pos.z * dp.plane() + p.pos.y * dp.pitch + p.pos.x * es;
    if (!p.unpack) { // domain -> buffer
      j.srcSlot = (const char *const *)(dom.dev_curr_slots() + p.qi);
      j.srcOff = domOff;
      j.srcPitch = dp.pitch;
      j.srcPlane = dp.plane();
      j.dstDirect = buf.ptr;
      j.dstOff = p.offset;
      j.dstPitch = rowBytes;
      j.dstPlane = rowBytes * p.ext.y;
    } else { // buffer -> domain
      j.srcDirect = buf.ptr;
      j.srcOff = p.offset;
      j.srcPitch = rowBytes;
      j.srcPlane = rowBytes * p.ext.y;
      j.dstSlot = (char *const *)(dom.dev_curr_slots() + p.qi);
      j.dstOff = domOff;
      j.dstPitch = dp.pitch;
      j.dstPlane = dp.plane();
    }
    const int w = pick_word(rowBytes, {j.srcOff, j.dstOff, j.srcPitch, j.dstPitch});
    j.wordBytes = w;
    j.extXw = (int32_t)(rowBytes / w);
    j.extY = (int32_t)p.ext.y;
    j.nWords = (int64_t)j.extXw * p.ext.y * p.ext.z;
    auto &batch = (p.unpack ? ub : pb)[{p.group, dom.gpu()}];
    batch.dev = dom.gpu();
    batch.jobs.push_back(j);
  }

  for (auto &kv : tb) translateBatches_[kv.first.first].push_back(std::move(kv.second));
  for (auto &kv : pb) packBatches_[kv.first.first].push_back(std::move(kv.second));
  for (auto &kv : ub) unpackBatches_[kv.first.first].push_back(std::move(kv.second));
}

hipStream_t ExchangeEngine::comm_stream_(int dev) {
  auto it = commStreams_.find(dev);
  if (it != commStreams_.end()) return it->second;
  STENCIL_HIP(hipSetDevice(dev));
  // NOTE: a high-priority stream here was measured 10% SLOWER on the
  // jacobi step (the preempting halo kernels serialized ahead of the
  // interior kernel instead of sharing the chip) -- keep default priority
  hipStream_t s;
  STENCIL_HIP(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  commStreams_[dev] = s;
  return s;
}

hipStream_t ExchangeEngine::pack_stream_(int dev) {
  auto it = packStreams_.find(dev);
  if (it != packStreams_.end()) return it->second;
  STENCIL_HIP(hipSetDevice(dev));
  hipStream_t s;
  STENCIL_HIP(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
  packStreams_[dev] = s;
  return s;
}

hipStream_t ExchangeEngine::compute_stream(int dom, int which) {
  auto &vec = which == 0   ? computeStreams_
              : which == 1 ? computeStreams2_
              : which == 2 ? computeStreams3_
                           : computeStreams4_;
  if (!vec[dom]) {
    STENCIL_HIP(hipSetDevice(domains_[dom]->gpu()));
    hipStream_t s;
    STENCIL_HIP(hipStreamCreateWithFlags(&s, hipStreamNonBlocking));
    vec[dom] = s;
  }
  return vec[dom];
}

// roctx ranges give rocprof-sys/rocprofv3 timelines the same phase
// annotations the reference had via NVTX (tracing)
void ExchangeEngine::launch_translates(int group) {
  roctxRangePush("stencil::translate");
  for (auto &b : translateBatches_[group]) b.launch(comm_stream_(b.dev));
  roctxRangePop();
}
void ExchangeEngine::launch_translates_plain_on(uintptr_t stream, int group) {
  for (auto &b : translateBatches_[group]) b.launch_plain((hipStream_t)stream);
}
void ExchangeEngine::launch_packs_plain_on(uintptr_t stream, int group) {
  for (auto &b : packBatches_[group]) b.launch_plain((hipStream_t)stream);
}
void ExchangeEngine::launch_unpacks_plain_on(uintptr_t stream, int group) {
  for (auto &b : unpackBatches_[group]) b.launch_plain((hipStream_t)stream);
}
void ExchangeEngine::launch_packs(int group) {
  roctxRangePush("stencil::pack");
  for (auto &b : packBatches_[group]) b.launch(pack_stream_(b.dev));
  roctxRangePop();
}
void ExchangeEngine::launch_unpacks(int group) {
  roctxRangePush("stencil::unpack");
  for (auto &b : unpackBatches_[group]) b.launch(pack_stream_(b.dev));
  roctxRangePop();
}

void ExchangeEngine::fence_packs_unpacks(int group) {
  // per-ENGINE events (a process-global map could be re-recorded by a
  // second engine between our record and wait)
  auto &evs = fenceEvents_;
  for (auto &b : packBatches_[group]) {
    if (b.jobs.empty()) continue;
    auto it = evs.find(b.dev);
    if (it == evs.end()) {
      STENCIL_HIP(hipSetDevice(b.dev));
      hipEvent_t e;
      STENCIL_HIP(hipEventCreateWithFlags(&e, hipEventDisableTiming));
      it = evs.emplace(b.dev, e).first;
    }
    STENCIL_HIP(hipSetDevice(b.dev));
    STENCIL_HIP(hipEventRecord(it->second, pack_stream_(b.dev)));
    for (auto &u : unpackBatches_[group]) {
      if (u.jobs.empty() || u.dev == b.dev) continue; // same stream: ordered
      STENCIL_HIP(hipSetDevice(u.dev));
      STENCIL_HIP(hipStreamWaitEvent(pack_stream_(u.dev), it->second, 0));
    }
  }
}

void ExchangeEngine::sync_translates() {
  roctxRangePush("stencil::sync_translates");
  for (auto &kv : commStreams_) {
    STENCIL_HIP(hipSetDevice(kv.first));
    STENCIL_HIP(hipStreamSynchronize(kv.second));
  }
  roctxRangePop();
}
void ExchangeEngine::sync_packs() {
  for (auto &kv : packStreams_) {
    STENCIL_HIP(hipSetDevice(kv.first));
    STENCIL_HIP(hipStreamSynchronize(kv.second));
  }
}
void ExchangeEngine::sync_all() {
  sync_translates();
  sync_packs();
}

void ExchangeEngine::sync_compute() {
  for (auto *vec : {&computeStreams_, &computeStreams2_, &computeStreams3_, &computeStreams4_})
    for (size_t i = 0; i < vec->size(); ++i)
      if ((*vec)[i]) {
        STENCIL_HIP(hipSetDevice(domains_[i]->gpu()));
        STENCIL_HIP(hipStreamSynchronize((*vec)[i]));
      }
}

} // namespace stencil_amd
