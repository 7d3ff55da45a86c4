// LocalDomain implementation (hipMalloc double buffers + device pointer
// tables). Reference behavior: src/local_domain.cu:159-220 (realize),
// :67-84 (swap), :86-129 (halo_pos), :131-157 (region_to_host).
#include "stencil_amd/domain.hpp"
#include "stencil_amd/hip_check.hpp"

#include <cstring>
#include <map>
#include <mutex>
#include <thread>
#include <stdexcept>

namespace stencil_amd {

LocalDomain::LocalDomain(const Vec3 &sz, const Vec3 &origin, int dev)
    : sz_(sz), origin_(origin), dev_(dev) {}

LocalDomain::~LocalDomain() {
  if (!realized_) return;
  (void)hipSetDevice(dev_);
  for (char *b : allocBases_)
    if (b) (void)hipFree(b);
  if (devCurrRaw_) (void)hipFree(devCurrRaw_);
  if (devNextRaw_) (void)hipFree(devNextRaw_);
}

int64_t LocalDomain::add_data(int64_t elemSize, const std::string &name) {
  elemSize_.push_back(elemSize);
  name_.push_back(name);
  return (int64_t)elemSize_.size() - 1;
}

void LocalDomain::realize() {
  if (realized_) throw std::runtime_error("LocalDomain::realize called twice");
  STENCIL_HIP(hipSetDevice(dev_));
  const Vec3 raw = raw_size();
  curr_.resize(elemSize_.size());
  next_.resize(elemSize_.size());
  padBytes_.resize(elemSize_.size(), 0);
  // the interior (overlap-mode) region starts at allocation x =
  // radius.x(-1) + max(-x-direction radii); pad the allocation so that
  // address is 16 B aligned and the vectorized stencil kernels have no
  // scalar head/tail cells
  int64_t shrink = 0;
  for (int dy = -1; dy <= 1; ++dy)
    for (int dz = -1; dz <= 1; ++dz) shrink = std::max(shrink, radius_.dir(-1, dy, dz));
  for (size_t qi = 0; qi < elemSize_.size(); ++qi) {
    const int64_t es = elemSize_[qi];
    // pitch rows to 256 B so every y-row starts on an HBM-friendly boundary
    const int64_t pitch = align_up(raw.x * es, 256);
    const int64_t interiorOff = (radius_.x(-1) + shrink) * es;
    const int64_t pad = (16 - interiorOff % 16) % 16;
    padBytes_[qi] = pad;
    const int64_t bytes = pitch * raw.y * raw.z + pad + 256;
    for (Pitched *buf : {&curr_[qi], &next_[qi]}) {
      buf->pitch = pitch;
      buf->ysize = raw.y;
      char *base = nullptr;
      STENCIL_HIP(hipMalloc((void **)&base, bytes));
      STENCIL_HIP(hipMemset(base, 0, bytes));
      allocBases_.push_back(base);
      buf->ptr = base + pad;
    }
  }
  // device pointer tables (fixed addresses; contents refreshed on swap)
  const int64_t n = num_data();
  STENCIL_HIP(hipMalloc((void **)&devCurrRaw_, n * sizeof(char *)));
  STENCIL_HIP(hipMalloc((void **)&devNextRaw_, n * sizeof(char *)));
  realized_ = true;
  swapUpload_();
}

void LocalDomain::swap() {
  std::swap(curr_, next_);
  swapUpload_();
}

namespace {
__global__ void swap_tables_kernel(char **a, char **b, int n) {
  const int i = threadIdx.x;
  if (i < n) {
    char *t = a[i];
    a[i] = b[i];
    b[i] = t;
  }
}
} // namespace

void LocalDomain::enqueue_table_swap(hipStream_t stream) {
  hipLaunchKernelGGL(swap_tables_kernel, dim3(1), dim3(256), 0, stream, devCurrRaw_, devNextRaw_,
                     (int)num_data());
  STENCIL_HIP(hipGetLastError());
}

void LocalDomain::swapUpload_() {
  STENCIL_HIP(hipSetDevice(dev_));
  const int64_t n = num_data();
  std::vector<char *> c(n), x(n);
  for (int64_t i = 0; i < n; ++i) {
    c[i] = curr_[i].ptr;
    x[i] = next_[i].ptr;
  }
  STENCIL_HIP(hipMemcpy(devCurrRaw_, c.data(), n * sizeof(char *), hipMemcpyHostToDevice));
  STENCIL_HIP(hipMemcpy(devNextRaw_, x.data(), n * sizeof(char *), hipMemcpyHostToDevice));
}

std::string LocalDomain::ipc_handle(int64_t qi, bool next) const {
  STENCIL_HIP(hipSetDevice(dev_));
  hipIpcMemHandle_t h;
  const Pitched &p = next ? next_.at(qi) : curr_.at(qi);
  // the handle must be taken on the allocation base (ptr is pad-offset);
  // importers add pad_bytes(qi)
  STENCIL_HIP(hipIpcGetMemHandle(&h, p.ptr - padBytes_.at(qi)));
  return std::string((const char *)&h, sizeof(h));
}

Vec3 LocalDomain::halo_pos(const Vec3 &dir, const Vec3 &sz, const Radius &radius, bool halo) {
  Vec3 ret;
  for (int i = 0; i < 3; ++i) {
    const int d = (int)dir[i];
    const int64_t rNeg = radius.dir(i == 0 ? -1 : 0, i == 1 ? -1 : 0, i == 2 ? -1 : 0);
    if (1 == d) {
      ret[i] = sz[i] + (halo ? rNeg : 0);
    } else if (-1 == d) {
      ret[i] = halo ? 0 : rNeg;
    } else {
      ret[i] = rNeg;
    }
  }
  return ret;
}

Rect3 LocalDomain::halo_coords(const Vec3 &dir, bool halo) const {
  Vec3 pos = halo_pos(dir, halo);
  const Vec3 ext = halo_extent(dir);
  // allocation coords -> global coords
  pos.x -= radius_.x(-1);
  pos.y -= radius_.y(-1);
  pos.z -= radius_.z(-1);
  pos += origin_;
  return Rect3(pos, pos + ext);
}

Rect3 LocalDomain::full_region() const {
  Vec3 lo = origin_;
  Vec3 hi = origin_ + sz_;
  lo.x -= radius_.x(-1);
  lo.y -= radius_.y(-1);
  lo.z -= radius_.z(-1);
  hi.x += radius_.x(1);
  hi.y += radius_.y(1);
  hi.z += radius_.z(1);
  return Rect3(lo, hi);
}

// Host I/O: one strided DMA per z-slab through a pinned double buffer
// (round 1 issued one hipMemcpy per ROW; a 3000^3 checkpoint took
// minutes). Pageable hipMemcpy3D measured 2.2 GB/s; the pinned bounce
// overlaps the DMA of slab k with the host memcpy of slab k-1.
namespace {
struct PinnedBounce {
  char *buf[2] = {nullptr, nullptr};
  int64_t cap = 0;
  hipStream_t stream = nullptr;
  hipEvent_t ev[2] = {nullptr, nullptr};
};
PinnedBounce &pinned_bounce(int dev, int64_t need) {
  static std::map<int, PinnedBounce> pools;
  static std::mutex mu;
  std::lock_guard<std::mutex> lk(mu);
  PinnedBounce &pp = pools[dev];
  constexpr int64_t kChunk = 32ll << 20;
  const int64_t want = std::max(need, kChunk);
  if (pp.cap < want) {
    for (int i = 0; i < 2; ++i) {
      if (pp.buf[i]) (void)hipHostFree(pp.buf[i]);
      STENCIL_HIP(hipHostMalloc((void **)&pp.buf[i], want));
    }
    pp.cap = want;
  }
  if (!pp.stream) {
    STENCIL_HIP(hipStreamCreateWithFlags(&pp.stream, hipStreamNonBlocking));
    for (int i = 0; i < 2; ++i)
      STENCIL_HIP(hipEventCreateWithFlags(&pp.ev[i], hipEventDisableTiming));
  }
  return pp;
}

// multi-threaded host memcpy: a single thread moves ~2.5 GB/s into fresh
// (page-faulting) numpy pages -- measured as the bottleneck of the whole
// pinned-bounce path (profiles/r2/r2_gpu7_io.log); 8 threads saturate the
// host memory system instead
void parallel_memcpy(void *dst, const void *src, int64_t bytes) {
  constexpr int64_t kMinPerThread = 4ll << 20;
  const int nt = (int)std::min<int64_t>(8, std::max<int64_t>(1, bytes / kMinPerThread));
  if (nt <= 1) {
    std::memcpy(dst, src, bytes);
    return;
  }
  std::vector<std::thread> ts;
  const int64_t per = (bytes + nt - 1) / nt;
  for (int i = 0; i < nt; ++i) {
    const int64_t off = i * per, n = std::min(per, bytes - off);
    if (n <= 0) break;
    ts.emplace_back([=]() { std::memcpy((char *)dst + off, (const char *)src + off, n); });
  }
  for (auto &t : ts) t.join();
}

// async strided copy of zs planes starting at zOff between the pitched
// device allocation and a contiguous host buffer
void slab_copy_async(const Pitched &p, int64_t es, const Vec3 &pos, const Vec3 &ext, int64_t zOff,
                     int64_t zs, void *host, bool toHost, hipStream_t stream) {
  const size_t rowBytes = (size_t)(ext.x * es);
  hipMemcpy3DParms prm{};
  const auto devPtr = make_hipPitchedPtr(p.ptr, (size_t)p.pitch, (size_t)p.pitch, (size_t)p.ysize);
  const auto devPos = make_hipPos((size_t)(pos.x * es), (size_t)pos.y, (size_t)(pos.z + zOff));
  const auto hostPtr = make_hipPitchedPtr(host, rowBytes, rowBytes, (size_t)ext.y);
  if (toHost) {
    prm.srcPtr = devPtr;
    prm.srcPos = devPos;
    prm.dstPtr = hostPtr;
    prm.kind = hipMemcpyDeviceToHost;
  } else {
    prm.srcPtr = hostPtr;
    prm.dstPtr = devPtr;
    prm.dstPos = devPos;
    prm.kind = hipMemcpyHostToDevice;
  }
  prm.extent = make_hipExtent(rowBytes, (size_t)ext.y, (size_t)zs);
  STENCIL_HIP(hipMemcpy3DAsync(&prm, stream));
}
} // namespace

void LocalDomain::region_to_host(void *dst, const Vec3 &pos, const Vec3 &ext, int64_t qi,
                                 bool fromNext) const {
  if (ext.x <= 0 || ext.y <= 0 || ext.z <= 0) return;
  STENCIL_HIP(hipSetDevice(dev_));
  const Pitched &p = fromNext ? next_[qi] : curr_[qi];
  const int64_t es = elemSize_[qi];
  const int64_t planeBytes = ext.x * es * ext.y;
  PinnedBounce &pp = pinned_bounce(dev_, planeBytes);
  const int64_t zPer = std::max<int64_t>(1, pp.cap / planeBytes);
  char *d = (char *)dst;
  int prev = -1;
  int64_t prevBytes = 0;
  int buf = 0;
  for (int64_t z0 = 0; z0 < ext.z; z0 += zPer, buf ^= 1) {
    const int64_t zs = std::min(zPer, ext.z - z0);
    slab_copy_async(p, es, pos, ext, z0, zs, pp.buf[buf], true, pp.stream);
    STENCIL_HIP(hipEventRecord(pp.ev[buf], pp.stream));
    if (prev >= 0) { // overlap the DMA with draining the previous slab
      parallel_memcpy(d, pp.buf[prev], prevBytes);
      d += prevBytes;
    }
    STENCIL_HIP(hipEventSynchronize(pp.ev[buf]));
    prev = buf;
    prevBytes = zs * planeBytes;
  }
  parallel_memcpy(d, pp.buf[prev], prevBytes);
}

void LocalDomain::region_from_host(const void *src, const Vec3 &pos, const Vec3 &ext, int64_t qi,
                                   bool toNext) const {
  if (ext.x <= 0 || ext.y <= 0 || ext.z <= 0) return;
  STENCIL_HIP(hipSetDevice(dev_));
  const Pitched &p = toNext ? next_[qi] : curr_[qi];
  const int64_t es = elemSize_[qi];
  const int64_t planeBytes = ext.x * es * ext.y;
  PinnedBounce &pp = pinned_bounce(dev_, planeBytes);
  const int64_t zPer = std::max<int64_t>(1, pp.cap / planeBytes);
  const char *s = (const char *)src;
  int buf = 0;
  for (int64_t z0 = 0; z0 < ext.z; z0 += zPer, buf ^= 1) {
    const int64_t zs = std::min(zPer, ext.z - z0);
    const int64_t bytes = zs * planeBytes;
    // the buffer's previous H2D (two slabs ago) must be complete
    STENCIL_HIP(hipEventSynchronize(pp.ev[buf]));
    parallel_memcpy(pp.buf[buf], s, bytes);
    s += bytes;
    slab_copy_async(p, es, pos, ext, z0, zs, pp.buf[buf], false, pp.stream);
    STENCIL_HIP(hipEventRecord(pp.ev[buf], pp.stream));
  }
  STENCIL_HIP(hipStreamSynchronize(pp.stream));
}

} // namespace stencil_amd
