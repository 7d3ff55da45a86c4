// Device-side field initialization kernels (reference: astaroth.cu:30-245
// sin/hash/const init kernels; ours: constant fill + harmonic modes, which
// give deterministic smooth initial conditions reproducible in NumPy).
#include <hip/hip_runtime.h>

#include "stencil_amd/accessor.hpp"
#include "stencil_amd/device_util.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"

namespace stencil_amd {

namespace {

struct InitParams {
  Accessor<double> acc; // grid-coordinate accessor (reference accessor.hpp)
  int64_t loX, loY, loZ;
  int32_t extX, extY, extZ;
  double base0, amp;
  double kx, ky, kz; // radians per cell
  double phase;
};

__global__ void __launch_bounds__(256) init_harmonic_f64_kernel(InitParams p) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz = blockIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const Vec3 g(p.loX + lx, p.loY + ly, p.loZ + lz);
  p.acc[g] = p.base0 + p.amp * sin(p.kx * g.x + p.ky * g.y + p.kz * g.z + p.phase);
}

// radial gaussian "explosion" (reference astaroth.cu:149-245
// radial_explosion_init_kernel: a velocity/field bump around a center):
// value = base + amp * exp(-r^2 / (2 sigma^2)), r = |g - center| in cells
struct RadialParams {
  Accessor<double> acc;
  int64_t loX, loY, loZ;
  int32_t extX, extY, extZ;
  double base0, amp;
  double cx, cy, cz;
  double inv2sigma2;
};

__global__ void __launch_bounds__(256) init_radial_f64_kernel(RadialParams p) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz = blockIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const Vec3 g(p.loX + lx, p.loY + ly, p.loZ + lz);
  const double dx = g.x - p.cx, dy = g.y - p.cy, dz = g.z - p.cz;
  p.acc[g] = p.base0 + p.amp * exp(-(dx * dx + dy * dy + dz * dz) * p.inv2sigma2);
}

} // namespace

void init_radial_f64(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, double base,
                     double amp, double cx, double cy, double cz, double sigma, bool nextBuf) {
  LocalDomain &d = eng.domain(dom);
  if (d.elem_size(qi) != 8) throw std::runtime_error("init_radial_f64: quantity must be fp64");
  const Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  RadialParams p{};
  p.acc = nextBuf ? next_accessor<double>(d, qi) : curr_accessor<double>(d, qi);
  p.loX = region.lo.x;
  p.loY = region.lo.y;
  p.loZ = region.lo.z;
  p.extX = (int32_t)ext.x;
  p.extY = (int32_t)ext.y;
  p.extZ = (int32_t)ext.z;
  p.base0 = base;
  p.amp = amp;
  p.cx = cx;
  p.cy = cy;
  p.cz = cz;
  p.inv2sigma2 = 1.0 / (2.0 * sigma * sigma);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  dim3 block(64, 4, 1);
  dim3 grid((uint32_t)((ext.x + 63) / 64), (uint32_t)((ext.y + 3) / 4), (uint32_t)ext.z);
  hipLaunchKernelGGL(init_radial_f64_kernel, grid, block, 0, eng.compute_stream(dom), p);
  STENCIL_HIP(hipGetLastError());
}

void init_harmonic_f64(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, double base,
                       double amp, double kx, double ky, double kz, double phase, bool nextBuf) {
  LocalDomain &d = eng.domain(dom);
  if (d.elem_size(qi) != 8) throw std::runtime_error("init_harmonic_f64: quantity must be fp64");
  const Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  InitParams p{};
  p.acc = nextBuf ? next_accessor<double>(d, qi) : curr_accessor<double>(d, qi);
  p.loX = region.lo.x;
  p.loY = region.lo.y;
  p.loZ = region.lo.z;
  p.extX = (int32_t)ext.x;
  p.extY = (int32_t)ext.y;
  p.extZ = (int32_t)ext.z;
  p.base0 = base;
  p.amp = amp;
  p.kx = kx;
  p.ky = ky;
  p.kz = kz;
  p.phase = phase;
  STENCIL_HIP(hipSetDevice(d.gpu()));
  dim3 block(64, 4, 1);
  dim3 grid((uint32_t)((ext.x + 63) / 64), (uint32_t)((ext.y + 3) / 4), (uint32_t)ext.z);
  hipLaunchKernelGGL(init_harmonic_f64_kernel, grid, block, 0, eng.compute_stream(dom), p);
  STENCIL_HIP(hipGetLastError());
}

} // namespace stencil_amd
