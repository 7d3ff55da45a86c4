// Field reductions: min / max / RMS of a quantity over a region
// (reference: astaroth/reductions.cuh). Per-block LDS tree reduction into
// a partials buffer, host-side finish (regions are O(1e8) cells; the
// partial array is a few thousand entries).
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cfloat>
#include <cmath>
#include <vector>

#include "stencil_amd/device_util.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"

namespace stencil_amd {

namespace {

struct RedParams {
  const char *base;
  int64_t pitch, plane;
  int64_t off; // byte offset of region start
  int32_t extX, extY, extZ;
  int32_t elemSize; // 4 or 8
  double *partials; // [nBlocks][3]: min, max, sumsq
};

__global__ void __launch_bounds__(256) reduce_kernel(RedParams p) {
  __shared__ double sMin[256], sMax[256], sSq[256];
  const char *base = p.base + p.off;
  const int64_t total = (int64_t)p.extX * p.extY * p.extZ;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  double mn = DBL_MAX, mx = -DBL_MAX, sq = 0.0;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const int32_t x = (int32_t)(i % p.extX);
    const int64_t t = i / p.extX;
    const int32_t y = (int32_t)(t % p.extY);
    const int64_t z = t / p.extY;
    const char *ptr = base + z * p.plane + (int64_t)y * p.pitch + (int64_t)x * p.elemSize;
    const double v = p.elemSize == 8 ? *(const double *)ptr : (double)*(const float *)ptr;
    mn = fmin(mn, v);
    mx = fmax(mx, v);
    sq += v * v;
  }
  const int tid = threadIdx.x;
  sMin[tid] = mn;
  sMax[tid] = mx;
  sSq[tid] = sq;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if (tid < s) {
      sMin[tid] = fmin(sMin[tid], sMin[tid + s]);
      sMax[tid] = fmax(sMax[tid], sMax[tid + s]);
      sSq[tid] += sSq[tid + s];
    }
    __syncthreads();
  }
  if (tid == 0) {
    p.partials[blockIdx.x * 3 + 0] = sMin[0];
    p.partials[blockIdx.x * 3 + 1] = sMax[0];
    p.partials[blockIdx.x * 3 + 2] = sSq[0];
  }
}

} // namespace

FieldStats field_stats(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                       bool nextBuf) {
  LocalDomain &d = eng.domain(dom);
  const Vec3 ext = region.extent();
  const int64_t total = ext.flatten();
  if (total <= 0) throw std::runtime_error("field_stats: empty region");
  const int64_t es = d.elem_size(qi);
  if (es != 4 && es != 8) throw std::runtime_error("field_stats: fp32/fp64 only");
  const Rect3 full = d.full_region();
  const Vec3 pos = region.lo - full.lo;
  const Pitched &pp = d.curr(qi);
  RedParams p{};
  p.base = nextBuf ? d.next(qi).ptr : d.curr(qi).ptr;
  p.pitch = pp.pitch;
  p.plane = pp.plane();
  p.off = pos.z * p.plane + pos.y * p.pitch + pos.x * es;
  p.extX = (int32_t)ext.x;
  p.extY = (int32_t)ext.y;
  p.extZ = (int32_t)ext.z;
  p.elemSize = (int32_t)es;
  const int blocks = (int)std::min<int64_t>((total + 255) / 256, 2048);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  STENCIL_HIP(hipMalloc((void **)&p.partials, blocks * 3 * sizeof(double)));
  hipLaunchKernelGGL(reduce_kernel, dim3(blocks), dim3(256), 0, eng.compute_stream(dom), p);
  STENCIL_HIP(hipGetLastError());
  std::vector<double> host(blocks * 3);
  STENCIL_HIP(hipStreamSynchronize(eng.compute_stream(dom)));
  STENCIL_HIP(
      hipMemcpy(host.data(), p.partials, blocks * 3 * sizeof(double), hipMemcpyDeviceToHost));
  STENCIL_HIP(hipFree(p.partials));
  FieldStats out{DBL_MAX, -DBL_MAX, 0.0};
  for (int b = 0; b < blocks; ++b) {
    out.min = std::min(out.min, host[b * 3 + 0]);
    out.max = std::max(out.max, host[b * 3 + 1]);
    out.rms += host[b * 3 + 2];
  }
  out.rms = std::sqrt(out.rms / (double)total);
  return out;
}

} // namespace stencil_amd
