// 6th-order compressible resistive MHD solver kernels (fp64, radius 3,
// 8 coupled fields) -- the Astaroth-class workload (reference:
// astaroth/astaroth.cu, user_kernels.h). This is an independent
// implementation of the standard equations, not a port of Astaroth's
// DSL-generated code.
//
// Fields: lnrho (log density), uu (velocity, 3), aa (magnetic vector
// potential, 3), ss (specific entropy). Equations (mu0 = 1, isothermal
// base state with entropy coupling; nu/eta/chi constant):
//   D lnrho / Dt = -div(u)
//   D u / Dt     = -cs2 * grad(lnrho + ss/cp) + j x B / rho
//                  + nu * (lap(u) + (1/3) grad(div(u)))
//   d A / dt     = u x B + eta * lap(A)          (resistive gauge)
//   D ss / Dt    = chi * lap(ss)
// with B = curl(A), j = grad(div(A)) - lap(A), D/Dt = d/dt + u . grad.
//
// Time integration: Williamson (1980) low-storage RK3 in the two-buffer
// form the halo-exchange library provides (curr/next + swap per substep):
//   next = curr + beta_s * (alpha_s / beta_{s-1} * (curr - next) + dt * rhs(curr))
//
// 6th-order central derivative coefficients; cross derivatives compose the
// first-derivative stencils (36-point quadrant sum).
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <stdexcept>

#include "stencil_amd/device_util.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"

namespace stencil_amd {

namespace {

struct MhdParams {
  // raw per-launch base pointers (kernarg > slot indirection; see jacobi)
  // quantities 0..7 = physics fields, 8 = div u, 9 = div A (auxiliary
  // exchanged fields of the separable-derivative scheme)
  const char *curr[10];
  char *next[10];
  int64_t pitch, plane; // byte strides (identical for all 8 fp64 fields)
  int64_t allocX, allocY, allocZ;
  int64_t loX, loY, loZ;
  int32_t extX, extY, extZ;
  double dsx, dsy, dsz; // grid spacing
  double dt;
  double cs2;    // sound speed squared
  double cp_inv; // 1/cp for the entropy pressure coupling
  double nu, eta, chi;
  double alpha_over_beta_prev; // alpha_s / beta_{s-1} (0 for substep 0)
  double beta;
  int32_t swizzle; // XCD-aware block remap (see xcd_remap)
  int32_t ychunk;  // y-chunked dispatch order in BLOCKS (see ychunk_remap)
  int32_t pad_;
};

enum { LNRHO = 0, UUX = 1, UUY = 2, UUZ = 3, AAX = 4, AAY = 5, AAZ = 6, SS = 7, DIVU = 8, DIVA = 9 };

// first derivative: (c1 (f1 - f-1) + c2 (f2 - f-2) + c3 (f3 - f-3)) / ds
__constant__ double D1[3] = {3.0 / 4.0, -3.0 / 20.0, 1.0 / 60.0};
// second derivative: (l0 f0 + sum li (fi + f-i)) / ds^2
__constant__ double D2[4] = {-49.0 / 18.0, 3.0 / 2.0, -3.0 / 20.0, 1.0 / 90.0};

struct Vec3d {
  double x, y, z;
};
__device__ inline Vec3d operator+(Vec3d a, Vec3d b) { return {a.x + b.x, a.y + b.y, a.z + b.z}; }
__device__ inline Vec3d operator-(Vec3d a, Vec3d b) { return {a.x - b.x, a.y - b.y, a.z - b.z}; }
__device__ inline Vec3d operator*(double s, Vec3d a) { return {s * a.x, s * a.y, s * a.z}; }
__device__ inline double dot(Vec3d a, Vec3d b) { return a.x * b.x + a.y * b.y + a.z * b.z; }
__device__ inline Vec3d cross(Vec3d a, Vec3d b) {
  return {a.y * b.z - a.z * b.y, a.z * b.x - a.x * b.z, a.x * b.y - a.y * b.x};
}

// read field q at offset (i,j,k) from the cell
struct Stencil {
  const char *base[10]; // per-field pointer AT the cell
  int64_t pitch, plane;

  __device__ double f(int q, int i, int j, int k) const {
    return *(const double *)(base[q] + (int64_t)k * plane + (int64_t)j * pitch + (int64_t)i * 8);
  }
  __device__ double c(int q) const { return *(const double *)base[q]; }

  __device__ double dx(int q, double ids) const {
    return (D1[0] * (f(q, 1, 0, 0) - f(q, -1, 0, 0)) + D1[1] * (f(q, 2, 0, 0) - f(q, -2, 0, 0)) +
            D1[2] * (f(q, 3, 0, 0) - f(q, -3, 0, 0))) *
           ids;
  }
  __device__ double dy(int q, double ids) const {
    return (D1[0] * (f(q, 0, 1, 0) - f(q, 0, -1, 0)) + D1[1] * (f(q, 0, 2, 0) - f(q, 0, -2, 0)) +
            D1[2] * (f(q, 0, 3, 0) - f(q, 0, -3, 0))) *
           ids;
  }
  __device__ double dz(int q, double ids) const {
    return (D1[0] * (f(q, 0, 0, 1) - f(q, 0, 0, -1)) + D1[1] * (f(q, 0, 0, 2) - f(q, 0, 0, -2)) +
            D1[2] * (f(q, 0, 0, 3) - f(q, 0, 0, -3))) *
           ids;
  }
  __device__ double dxx(int q, double ids2) const {
    return (D2[0] * c(q) + D2[1] * (f(q, 1, 0, 0) + f(q, -1, 0, 0)) +
            D2[2] * (f(q, 2, 0, 0) + f(q, -2, 0, 0)) + D2[3] * (f(q, 3, 0, 0) + f(q, -3, 0, 0))) *
           ids2;
  }
  __device__ double dyy(int q, double ids2) const {
    return (D2[0] * c(q) + D2[1] * (f(q, 0, 1, 0) + f(q, 0, -1, 0)) +
            D2[2] * (f(q, 0, 2, 0) + f(q, 0, -2, 0)) + D2[3] * (f(q, 0, 3, 0) + f(q, 0, -3, 0))) *
           ids2;
  }
  __device__ double dzz(int q, double ids2) const {
    return (D2[0] * c(q) + D2[1] * (f(q, 0, 0, 1) + f(q, 0, 0, -1)) +
            D2[2] * (f(q, 0, 0, 2) + f(q, 0, 0, -2)) + D2[3] * (f(q, 0, 0, 3) + f(q, 0, 0, -3))) *
           ids2;
  }
  // cross derivatives: composed first-derivative stencils (quadrant sum)
  __device__ double dxy(int q, double idsx, double idsy) const {
    double s = 0;
#pragma unroll
    for (int i = 1; i <= 3; ++i)
#pragma unroll
      for (int j = 1; j <= 3; ++j)
        s += D1[i - 1] * D1[j - 1] *
             (f(q, i, j, 0) - f(q, i, -j, 0) - f(q, -i, j, 0) + f(q, -i, -j, 0));
    return s * idsx * idsy;
  }
  __device__ double dxz(int q, double idsx, double idsz) const {
    double s = 0;
#pragma unroll
    for (int i = 1; i <= 3; ++i)
#pragma unroll
      for (int k = 1; k <= 3; ++k)
        s += D1[i - 1] * D1[k - 1] *
             (f(q, i, 0, k) - f(q, i, 0, -k) - f(q, -i, 0, k) + f(q, -i, 0, -k));
    return s * idsx * idsz;
  }
  __device__ double dyz(int q, double idsy, double idsz) const {
    double s = 0;
#pragma unroll
    for (int j = 1; j <= 3; ++j)
#pragma unroll
      for (int k = 1; k <= 3; ++k)
        s += D1[j - 1] * D1[k - 1] *
             (f(q, 0, j, k) - f(q, 0, j, -k) - f(q, 0, -j, k) + f(q, 0, -j, -k));
    return s * idsy * idsz;
  }

  __device__ Vec3d grad(int q, double ix, double iy, double iz) const {
    return {dx(q, ix), dy(q, iy), dz(q, iz)};
  }
  __device__ double lap(int q, double ix, double iy, double iz) const {
    return dxx(q, ix * ix) + dyy(q, iy * iy) + dzz(q, iz * iz);
  }
};

// The solver is split into two kernels per substep: the monolithic form
// allocated 356 VGPR+AGPR (1 wave/SIMD) and latency-bound on its dependent
// cache loads. Split, each kernel fits >=2 waves/SIMD; the duplicated
// center reads are served by L1/L2.

struct MhdCommon {
  const char *base[10];
  int64_t pitch, plane;
  char *out[10];
};

// XCD-aware block remap (STENCIL_MHD_SWIZZLE=1): gives each XCD a
// contiguous slab of the block grid. MEASURED 10% SLOWER on the MHD
// kernels (10.05 -> 11.12 ms/iter, reproducible within-box) despite
// their 50.8% L2 hit-rate (profiles/): the default round-robin dispatch
// evidently spreads HBM channel traffic better than slab locality helps.
// Kept off by default as an experiment hook.
__device__ __forceinline__ void xcd_remap(int32_t &bx, int32_t &by, int32_t &bz) {
  const int32_t nwg = gridDim.x * gridDim.y * gridDim.z;
  const int32_t flat = bx + gridDim.x * (by + gridDim.y * bz);
  const int32_t q = nwg / 8, r = nwg % 8;
  const int32_t xcd = flat % 8, idx = flat / 8;
  const int32_t nf = xcd < r ? xcd * (q + 1) + idx : r * (q + 1) + (xcd - r) * q + idx;
  bx = nf % gridDim.x;
  by = (nf / gridDim.x) % gridDim.y;
  bz = nf / (gridDim.x * gridDim.y);
}

// y-chunked dispatch (STENCIL_MHD_YCHUNK=<blocks>): iterate the block
// grid chunk-of-y-rows-major (for each y-chunk: all z, then y-in-chunk,
// then x) so a z-slab's working set per XCD shrinks by nby/C -- the
// momentum/scalar kernels refetch z-star planes ~3.4x because a 2-thick
// slab of 10 fp64 fields (~5 MB/XCD at 256^3) thrashes the 4 MB XCD L2.
// C must divide gridDim.y (host guarantees).
__device__ __forceinline__ void ychunk_remap(int32_t C, int32_t &bx, int32_t &by, int32_t &bz) {
  const int32_t flat = bx + gridDim.x * (by + gridDim.y * bz);
  const int32_t per = gridDim.x * C * gridDim.z;
  const int32_t chunk = flat / per, rem = flat % per;
  bz = rem / (gridDim.x * C);
  const int32_t rem2 = rem % (gridDim.x * C);
  by = chunk * C + rem2 / gridDim.x;
  bx = rem2 % gridDim.x;
}

__device__ __forceinline__ MhdCommon mhd_setup(const MhdParams &p, int32_t lx, int32_t ly,
                                               int32_t lz) {
  const int64_t ax = p.loX + lx - p.allocX;
  const int64_t ay = p.loY + ly - p.allocY;
  const int64_t az = p.loZ + lz - p.allocZ;
  const int64_t cellOff = az * p.plane + ay * p.pitch + ax * 8;
  MhdCommon c;
  c.pitch = p.pitch;
  c.plane = p.plane;
#pragma unroll
  for (int q = 0; q < 10; ++q) {
    c.base[q] = p.curr[q] + cellOff;
    c.out[q] = p.next[q] + cellOff;
  }
  return c;
}

__device__ __forceinline__ void write_rk3(const MhdParams &p, const Stencil &st, char *out, int q,
                                          double r) {
  const double cur = st.c(q);
  const double prev = *(const double *)out;
  *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) + p.dt * r);
}

// kernel 1: continuity + entropy + induction (lnrho, ss, aa). First and
// second derivatives only, no cross terms.
__device__ __forceinline__ void mhd_scalar_body(const MhdParams &p) {
  int32_t bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  if (p.swizzle) xcd_remap(bx, by, bz);
  if (p.ychunk) ychunk_remap(p.ychunk, bx, by, bz);
  const int32_t lx = bx * blockDim.x + threadIdx.x;
  const int32_t ly = by * blockDim.y + threadIdx.y;
  const int32_t lz = bz * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 10; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const Vec3d uu = {st.c(UUX), st.c(UUY), st.c(UUZ)};
  {
    const Vec3d glnrho = st.grad(LNRHO, ix, iy, iz);
    write_rk3(p, st, c.out[LNRHO], LNRHO, -dot(uu, glnrho) - st.c(DIVU));
  }
  {
    const Vec3d gss = st.grad(SS, ix, iy, iz);
    write_rk3(p, st, c.out[SS], SS, -dot(uu, gss) + p.chi * st.lap(SS, ix, iy, iz));
  }
  const Vec3d B = {st.dy(AAZ, iy) - st.dz(AAY, iz), st.dz(AAX, iz) - st.dx(AAZ, ix),
                   st.dx(AAY, ix) - st.dy(AAX, iy)};
  const Vec3d uxB = cross(uu, B);
  write_rk3(p, st, c.out[AAX], AAX, uxB.x + p.eta * st.lap(AAX, ix, iy, iz));
  write_rk3(p, st, c.out[AAY], AAY, uxB.y + p.eta * st.lap(AAY, ix, iy, iz));
  write_rk3(p, st, c.out[AAZ], AAZ, uxB.z + p.eta * st.lap(AAZ, ix, iy, iz));
}
__global__ void __launch_bounds__(256) mhd_scalar_kernel(MhdParams p) { mhd_scalar_body(p); }
// 512-thread variant (STENCIL_MHD_BLOCK with 512 threads, e.g. 64x4x2):
// a taller block halves the per-cell refetch of +-3 y-rows and z-planes
// through L1/LDS-free sharing -- the MHD kernels are L2-bandwidth-bound
// (docs/DESIGN.md round-2 analysis), so block-footprint reuse is the lever
__global__ void __launch_bounds__(512) mhd_scalar_kernel_b512(MhdParams p) { mhd_scalar_body(p); }

// kernel 2a: Lorentz force j x B into a scratch array (j needs the cross
// derivatives of A -- the most load-heavy part of the solver; isolating it
// keeps each kernel's register set small enough for >=2 waves/SIMD)
// div pass: divu = div(u), divA = div(A), stored as exchanged auxiliary
// quantities. After their halos are exchanged, grad(div .) becomes three
// cheap 7-point first derivatives of ONE field instead of 36-point
// composed cross-derivative sums (separable formulation; identical order
// of accuracy, fp-rounding-level difference mirrored exactly in the NumPy
// reference).
__device__ __forceinline__ void mhd_div_body(const MhdParams &p) {
  int32_t bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  if (p.swizzle) xcd_remap(bx, by, bz);
  if (p.ychunk) ychunk_remap(p.ychunk, bx, by, bz);
  const int32_t lx = bx * blockDim.x + threadIdx.x;
  const int32_t ly = by * blockDim.y + threadIdx.y;
  const int32_t lz = bz * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 10; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  // div fields live in the CURR buffers (consumed this substep, stale
  // after swap, recomputed next substep)
  *(double *)(const_cast<char *>(c.base[DIVU])) =
      st.dx(UUX, ix) + st.dy(UUY, iy) + st.dz(UUZ, iz);
  *(double *)(const_cast<char *>(c.base[DIVA])) =
      st.dx(AAX, ix) + st.dy(AAY, iy) + st.dz(AAZ, iz);
}
__global__ void __launch_bounds__(256) mhd_div_kernel(MhdParams p) { mhd_div_body(p); }
__global__ void __launch_bounds__(512) mhd_div_kernel_b512(MhdParams p) { mhd_div_body(p); }

// momentum: j_i = D_i(divA) - lap(A_i); graddiv u = grad(divA... grad(divu)
__device__ __forceinline__ void mhd_momentum_body(const MhdParams &p) {
  int32_t bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  if (p.swizzle) xcd_remap(bx, by, bz);
  if (p.ychunk) ychunk_remap(p.ychunk, bx, by, bz);
  const int32_t lx = bx * blockDim.x + threadIdx.x;
  const int32_t ly = by * blockDim.y + threadIdx.y;
  const int32_t lz = bz * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 10; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const Vec3d uu = {st.c(UUX), st.c(UUY), st.c(UUZ)};
  const double rho_inv = exp(-st.c(LNRHO));

  const Vec3d j = {st.dx(DIVA, ix) - st.lap(AAX, ix, iy, iz),
                   st.dy(DIVA, iy) - st.lap(AAY, ix, iy, iz),
                   st.dz(DIVA, iz) - st.lap(AAZ, ix, iy, iz)};
  const Vec3d B = {st.dy(AAZ, iy) - st.dz(AAY, iz), st.dz(AAX, iz) - st.dx(AAZ, ix),
                   st.dx(AAY, ix) - st.dy(AAX, iy)};
  const Vec3d jxB = cross(j, B);

  {
    const double ugradu = uu.x * st.dx(UUX, ix) + uu.y * st.dy(UUX, iy) + uu.z * st.dz(UUX, iz);
    const double press = st.dx(LNRHO, ix) + p.cp_inv * st.dx(SS, ix);
    const double visc = p.nu * (st.lap(UUX, ix, iy, iz) + st.dx(DIVU, ix) / 3.0);
    write_rk3(p, st, c.out[UUX], UUX, -ugradu - p.cs2 * press + rho_inv * jxB.x + visc);
  }
  {
    const double ugradu = uu.x * st.dx(UUY, ix) + uu.y * st.dy(UUY, iy) + uu.z * st.dz(UUY, iz);
    const double press = st.dy(LNRHO, iy) + p.cp_inv * st.dy(SS, iy);
    const double visc = p.nu * (st.lap(UUY, ix, iy, iz) + st.dy(DIVU, iy) / 3.0);
    write_rk3(p, st, c.out[UUY], UUY, -ugradu - p.cs2 * press + rho_inv * jxB.y + visc);
  }
  {
    const double ugradu = uu.x * st.dx(UUZ, ix) + uu.y * st.dy(UUZ, iy) + uu.z * st.dz(UUZ, iz);
    const double press = st.dz(LNRHO, iz) + p.cp_inv * st.dz(SS, iz);
    const double visc = p.nu * (st.lap(UUZ, ix, iy, iz) + st.dz(DIVU, iz) / 3.0);
    write_rk3(p, st, c.out[UUZ], UUZ, -ugradu - p.cs2 * press + rho_inv * jxB.z + visc);
  }
}

__global__ void __launch_bounds__(256) mhd_momentum_kernel(MhdParams p) { mhd_momentum_body(p); }

// Per-component momentum (STENCIL_MHD_MOMSPLIT=1): three kernels, one
// velocity component each, on three CONCURRENT streams. Each kernel
// needs only 2 of 3 j and B components (jxB_c = j_a B_b - j_b B_a), so
// total work rises ~2x on the shared j/B part -- but chip-wide
// concurrent wave count triples, the one axis the per-wave experiments
// (z-march, z-pair, 512-blocks) could not move. Counter-guided: the
// kernels are 83% SQ_WAIT stalled, so extra independent waves fill
// latency the way extra per-wave ILP measurably did not.
template <int C>
__global__ void __launch_bounds__(256) mhd_momentum_comp_kernel(MhdParams p) {
  int32_t bx = blockIdx.x, by = blockIdx.y, bz = blockIdx.z;
  if (p.ychunk) ychunk_remap(p.ychunk, bx, by, bz);
  const int32_t lx = bx * blockDim.x + threadIdx.x;
  const int32_t ly = by * blockDim.y + threadIdx.y;
  const int32_t lz = bz * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 10; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const double ids1[3] = {ix, iy, iz};

  // axis-generic first derivative / laplacian (axis folds at compile time)
  auto d1 = [&](int q, int axis) {
    return axis == 0 ? st.dx(q, ix) : axis == 1 ? st.dy(q, iy) : st.dz(q, iz);
  };
  auto lap = [&](int q) { return st.lap(q, ix, iy, iz); };

  constexpr int A = (C + 1) % 3, Bc = (C + 2) % 3;
  // j_k = d_k(divA) - lap(A_k);  B_k = d_{k+1}(A_{k+2}) - d_{k+2}(A_{k+1})
  const double jA = d1(DIVA, A) - lap(AAX + A);
  const double jB = d1(DIVA, Bc) - lap(AAX + Bc);
  const double BA = d1(AAX + (A + 2) % 3, (A + 1) % 3) - d1(AAX + (A + 1) % 3, (A + 2) % 3);
  const double BB = d1(AAX + (Bc + 2) % 3, (Bc + 1) % 3) - d1(AAX + (Bc + 1) % 3, (Bc + 2) % 3);
  const double jxB_c = jA * BB - jB * BA;

  const double rho_inv = exp(-st.c(LNRHO));
  const double ugradu =
      st.c(UUX) * st.dx(UUX + C, ix) + st.c(UUY) * st.dy(UUX + C, iy) + st.c(UUZ) * st.dz(UUX + C, iz);
  const double press = d1(LNRHO, C) + p.cp_inv * d1(SS, C);
  const double visc = p.nu * (lap(UUX + C) + d1(DIVU, C) / 3.0);
  (void)ids1;
  write_rk3(p, st, c.out[UUX + C], UUX + C, -ugradu - p.cs2 * press + rho_inv * jxB_c + visc);
}

// z-pair momentum (STENCIL_MHD_MOM2=1): each thread computes cells z and
// z+1. Counter-guided experiment: the kernels are 83% SQ_WAIT stalled
// (profiles/astaroth_256_sq_counters_r2.csv), so (a) two independent
// dependency chains per wave double the loads in flight, and (b) the two
// cells SHARE their z-star -- the 7-deep center columns overlap in 6 of
// 8 values per field, cutting the L2-missing z-displaced loads ~43%.
// In-plane derivatives still come from L1/L2 via the Stencil helper.
struct ZPair {
  double v[8]; // f(q, 0, 0, z-3 .. z+4)
  __device__ __forceinline__ double dz0(double ids) const {
    return (D1[0] * (v[4] - v[2]) + D1[1] * (v[5] - v[1]) + D1[2] * (v[6] - v[0])) * ids;
  }
  __device__ __forceinline__ double dz1(double ids) const {
    return (D1[0] * (v[5] - v[3]) + D1[1] * (v[6] - v[2]) + D1[2] * (v[7] - v[1])) * ids;
  }
  __device__ __forceinline__ double dzz0(double ids2) const {
    return (D2[0] * v[3] + D2[1] * (v[4] + v[2]) + D2[2] * (v[5] + v[1]) + D2[3] * (v[6] + v[0])) *
           ids2;
  }
  __device__ __forceinline__ double dzz1(double ids2) const {
    return (D2[0] * v[4] + D2[1] * (v[5] + v[3]) + D2[2] * (v[6] + v[2]) + D2[3] * (v[7] + v[1])) *
           ids2;
  }
  __device__ __forceinline__ double c0() const { return v[3]; }
  __device__ __forceinline__ double c1() const { return v[4]; }
};

__global__ void __launch_bounds__(256) mhd_momentum_pair_kernel(MhdParams p) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz0 = 2 * (blockIdx.z * blockDim.z + threadIdx.z);
  if (lx >= p.extX || ly >= p.extY || lz0 >= p.extZ) return;
  const bool two = lz0 + 1 < p.extZ;
  const MhdCommon c = mhd_setup(p, lx, ly, lz0);
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const int64_t plane = p.plane;

  ZPair zp[10];
#pragma unroll
  for (int q = 0; q < 10; ++q)
#pragma unroll
    for (int k = 0; k < 8; ++k)
      zp[q].v[k] = *(const double *)(c.base[q] + (int64_t)(k - 3) * plane);

  Stencil st0, st1; // in-plane helpers for the two z layers
  st0.pitch = st1.pitch = p.pitch;
  st0.plane = st1.plane = p.plane;
#pragma unroll
  for (int q = 0; q < 10; ++q) {
    st0.base[q] = c.base[q];
    st1.base[q] = c.base[q] + plane;
  }

#pragma unroll
  for (int cell = 0; cell < 2; ++cell) {
    if (cell == 1 && !two) break;
    const Stencil &st = cell ? st1 : st0;
    const Vec3d uu = cell ? Vec3d{zp[UUX].c1(), zp[UUY].c1(), zp[UUZ].c1()}
                          : Vec3d{zp[UUX].c0(), zp[UUY].c0(), zp[UUZ].c0()};
    const double rho_inv = exp(-(cell ? zp[LNRHO].c1() : zp[LNRHO].c0()));
    auto DZ = [&](int q) { return cell ? zp[q].dz1(iz) : zp[q].dz0(iz); };
    auto DZZ = [&](int q) { return cell ? zp[q].dzz1(iz * iz) : zp[q].dzz0(iz * iz); };
    auto LAP = [&](int q) { return st.dxx(q, ix * ix) + st.dyy(q, iy * iy) + DZZ(q); };
    const Vec3d j = {st.dx(DIVA, ix) - LAP(AAX), st.dy(DIVA, iy) - LAP(AAY),
                     DZ(DIVA) - LAP(AAZ)};
    const Vec3d B = {st.dy(AAZ, iy) - DZ(AAY), DZ(AAX) - st.dx(AAZ, ix),
                     st.dx(AAY, ix) - st.dy(AAX, iy)};
    const Vec3d jxB = cross(j, B);
    const int64_t outOff = (int64_t)cell * plane;
    {
      const double ugradu = uu.x * st.dx(UUX, ix) + uu.y * st.dy(UUX, iy) + uu.z * DZ(UUX);
      const double press = st.dx(LNRHO, ix) + p.cp_inv * st.dx(SS, ix);
      const double visc = p.nu * (LAP(UUX) + st.dx(DIVU, ix) / 3.0);
      char *out = c.out[UUX] + outOff;
      const double cur = cell ? zp[UUX].c1() : zp[UUX].c0();
      const double prev = *(const double *)out;
      *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) +
                                       p.dt * (-ugradu - p.cs2 * press + rho_inv * jxB.x + visc));
    }
    {
      const double ugradu = uu.x * st.dx(UUY, ix) + uu.y * st.dy(UUY, iy) + uu.z * DZ(UUY);
      const double press = st.dy(LNRHO, iy) + p.cp_inv * st.dy(SS, iy);
      const double visc = p.nu * (LAP(UUY) + st.dy(DIVU, iy) / 3.0);
      char *out = c.out[UUY] + outOff;
      const double cur = cell ? zp[UUY].c1() : zp[UUY].c0();
      const double prev = *(const double *)out;
      *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) +
                                       p.dt * (-ugradu - p.cs2 * press + rho_inv * jxB.y + visc));
    }
    {
      const double ugradu = uu.x * st.dx(UUZ, ix) + uu.y * st.dy(UUZ, iy) + uu.z * DZ(UUZ);
      const double press = DZ(LNRHO) + p.cp_inv * DZ(SS);
      const double visc = p.nu * (LAP(UUZ) + DZ(DIVU) / 3.0);
      char *out = c.out[UUZ] + outOff;
      const double cur = cell ? zp[UUZ].c1() : zp[UUZ].c0();
      const double prev = *(const double *)out;
      *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) +
                                       p.dt * (-ugradu - p.cs2 * press + rho_inv * jxB.z + visc));
    }
  }
}
__global__ void __launch_bounds__(512) mhd_momentum_kernel_b512(MhdParams p) {
  mhd_momentum_body(p);
}
// occupancy experiment (STENCIL_MHD_MOM5=1): force 5 waves/SIMD -- the
// compiler must fit ~102 VGPR, possibly spilling; measured A/B decides
__global__ void __launch_bounds__(256, 5) mhd_momentum_kernel_w5(MhdParams p) {
  mhd_momentum_body(p);
}

// z-marching momentum (STENCIL_MHD_ZMARCH=1): each thread walks a whole
// z-column keeping a 7-deep register pipeline of every field's center
// column. The separable scheme (div u / div A exchanged as fields) left
// momentum with NO cross derivatives, so its z-direction reads are
// exactly the center-column values f(q,0,0,+-1..3) -- the 60 loads/cell
// of z-displaced planes that thrash the 4 MB per-XCD L2 (the measured
// 3.4x per-field fetch ratio, profiles/astaroth_256_fetch_size.csv).
// The pipeline turns ALL of them into register shifts; what remains are
// in-plane x/y-offset loads (served by L1/L2 plane locality) plus ONE
// new z+3 load per field per step. Cost: ~70 fp64 pipeline registers ->
// 1-2 waves/SIMD; an HBM-bound kernel can afford low occupancy if the
// per-wave load stream stays deep.
struct ZPipe {
  double v[7]; // v[k] = f(q, 0, 0, k-3)
  __device__ __forceinline__ void shift(double next) {
#pragma unroll
    for (int i = 0; i < 6; ++i) v[i] = v[i + 1];
    v[6] = next;
  }
  __device__ __forceinline__ double dz(double ids) const {
    return (D1[0] * (v[4] - v[2]) + D1[1] * (v[5] - v[1]) + D1[2] * (v[6] - v[0])) * ids;
  }
  __device__ __forceinline__ double dzz(double ids2) const {
    return (D2[0] * v[3] + D2[1] * (v[4] + v[2]) + D2[2] * (v[5] + v[1]) + D2[3] * (v[6] + v[0])) *
           ids2;
  }
  __device__ __forceinline__ double c() const { return v[3]; }
};

__global__ void __launch_bounds__(128) mhd_momentum_zmarch_kernel(MhdParams p) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  if (lx >= p.extX || ly >= p.extY) return;
  const MhdCommon c0 = mhd_setup(p, lx, ly, 0);
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const int64_t plane = p.plane;

  ZPipe zp[10];
#pragma unroll
  for (int q = 0; q < 10; ++q)
#pragma unroll
    for (int k = 0; k < 7; ++k)
      zp[q].v[k] = *(const double *)(c0.base[q] + (int64_t)(k - 3) * plane);

  Stencil st; // in-plane derivative helper; base advanced per z step
  st.pitch = p.pitch;
  st.plane = p.plane;

  for (int32_t lz = 0;;) {
#pragma unroll
    for (int q = 0; q < 10; ++q) st.base[q] = c0.base[q] + (int64_t)lz * plane;
    const Vec3d uu = {zp[UUX].c(), zp[UUY].c(), zp[UUZ].c()};
    const double rho_inv = exp(-zp[LNRHO].c());
    const Vec3d j = {st.dx(DIVA, ix) - (st.dxx(AAX, ix * ix) + st.dyy(AAX, iy * iy) + zp[AAX].dzz(iz * iz)),
                     st.dy(DIVA, iy) - (st.dxx(AAY, ix * ix) + st.dyy(AAY, iy * iy) + zp[AAY].dzz(iz * iz)),
                     zp[DIVA].dz(iz) - (st.dxx(AAZ, ix * ix) + st.dyy(AAZ, iy * iy) + zp[AAZ].dzz(iz * iz))};
    const Vec3d B = {st.dy(AAZ, iy) - zp[AAY].dz(iz), zp[AAX].dz(iz) - st.dx(AAZ, ix),
                     st.dx(AAY, ix) - st.dy(AAX, iy)};
    const Vec3d jxB = cross(j, B);
    {
      const double ugradu = uu.x * st.dx(UUX, ix) + uu.y * st.dy(UUX, iy) + uu.z * zp[UUX].dz(iz);
      const double press = st.dx(LNRHO, ix) + p.cp_inv * st.dx(SS, ix);
      const double visc =
          p.nu * (st.dxx(UUX, ix * ix) + st.dyy(UUX, iy * iy) + zp[UUX].dzz(iz * iz) +
                  st.dx(DIVU, ix) / 3.0);
      char *out = c0.out[UUX] + (int64_t)lz * plane;
      const double cur = zp[UUX].c(), prev = *(const double *)out;
      *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) +
                                       p.dt * (-ugradu - p.cs2 * press + rho_inv * jxB.x + visc));
    }
    {
      const double ugradu = uu.x * st.dx(UUY, ix) + uu.y * st.dy(UUY, iy) + uu.z * zp[UUY].dz(iz);
      const double press = st.dy(LNRHO, iy) + p.cp_inv * st.dy(SS, iy);
      const double visc =
          p.nu * (st.dxx(UUY, ix * ix) + st.dyy(UUY, iy * iy) + zp[UUY].dzz(iz * iz) +
                  st.dy(DIVU, iy) / 3.0);
      char *out = c0.out[UUY] + (int64_t)lz * plane;
      const double cur = zp[UUY].c(), prev = *(const double *)out;
      *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) +
                                       p.dt * (-ugradu - p.cs2 * press + rho_inv * jxB.y + visc));
    }
    {
      const double ugradu = uu.x * st.dx(UUZ, ix) + uu.y * st.dy(UUZ, iy) + uu.z * zp[UUZ].dz(iz);
      const double press = zp[LNRHO].dz(iz) + p.cp_inv * zp[SS].dz(iz);
      const double visc =
          p.nu * (st.dxx(UUZ, ix * ix) + st.dyy(UUZ, iy * iy) + zp[UUZ].dzz(iz * iz) +
                  zp[DIVU].dz(iz) / 3.0);
      char *out = c0.out[UUZ] + (int64_t)lz * plane;
      const double cur = zp[UUZ].c(), prev = *(const double *)out;
      *(double *)out = cur + p.beta * (p.alpha_over_beta_prev * (cur - prev) +
                                       p.dt * (-ugradu - p.cs2 * press + rho_inv * jxB.z + visc));
    }
    if (++lz >= p.extZ) break;
#pragma unroll
    for (int q = 0; q < 10; ++q)
      zp[q].shift(*(const double *)(c0.base[q] + (int64_t)(lz + 3) * plane));
  }
}

} // namespace

static void mhd_fill_params(LocalDomain &d, const Rect3 &region, const MhdCoeffs &cf,
                            MhdParams &p) {
  if (d.num_data() != 10 || d.elem_size(0) != 8)
    throw std::runtime_error("mhd: domain must have 10 fp64 quantities (8 fields + divu + divA)");
  for (int q = 0; q < 10; ++q) {
    p.curr[q] = d.curr(q).ptr;
    p.next[q] = d.next(q).ptr;
    if (d.curr(q).pitch != d.curr(0).pitch) throw std::runtime_error("mhd: field pitches differ");
  }
  p.pitch = d.curr(0).pitch;
  p.plane = d.curr(0).plane();
  const Rect3 full = d.full_region();
  p.allocX = full.lo.x;
  p.allocY = full.lo.y;
  p.allocZ = full.lo.z;
  p.loX = region.lo.x;
  p.loY = region.lo.y;
  p.loZ = region.lo.z;
  const Vec3 ext = region.extent();
  p.extX = (int32_t)ext.x;
  p.extY = (int32_t)ext.y;
  p.extZ = (int32_t)ext.z;
  p.dsx = cf.dsx;
  p.dsy = cf.dsy;
  p.dsz = cf.dsz;
  p.cs2 = cf.cs2;
  p.cp_inv = cf.cp_inv;
  p.nu = cf.nu;
  p.eta = cf.eta;
  p.chi = cf.chi;
  static int swz = -1;
  if (swz < 0) {
    const char *e = getenv("STENCIL_MHD_SWIZZLE");
    swz = (e && e[0] == '1') ? 1 : 0;
  }
  p.swizzle = swz;
  p.ychunk = 0; // set per-launch (needs the real grid) in the launchers
}

// STENCIL_MHD_YCHUNK=<blocks> (0 = off; default 32): largest divisor of
// the y-block count that is <= the request (C must divide gridDim.y for
// the in-kernel decode). Measured: +1.5% at 256^3, +21% at 640^3 (fixes
// the odd-size L2-thrash anomaly with the default block shape); the
// earlier whole-grid xcd_remap failure does not apply -- this keeps the
// round-robin XCD spread WITHIN a chunk while bounding the z-slab
// working set that the chunk streams through the XCD L2s.
static int32_t mhd_ychunk(int32_t nby) {
  static int ych = -1;
  if (ych < 0) {
    const char *e = getenv("STENCIL_MHD_YCHUNK");
    ych = e ? atoi(e) : 32;
  }
  if (ych <= 0) return 0;
  for (int32_t c = ych < nby ? ych : nby; c >= 1; --c)
    if (nby % c == 0) return c == nby ? 0 : c;
  return 0;
}

static dim3 mhd_block() {
  // 64x2x2 measured best after the separable-derivative restructure
  // (sweep in gpurun gpu15 log: 1649 vs 1608 Mcell/s at 32x4x2)
  static int bx = 0, by = 0, bz = 0;
  if (!bx) {
    bx = 64;
    by = 2;
    bz = 2;
    if (const char *e = getenv("STENCIL_MHD_BLOCK"))
      if (sscanf(e, "%dx%dx%d", &bx, &by, &bz) != 3 ||
          (bx * by * bz != 256 && bx * by * bz != 512)) {
        bx = 64;
        by = 2;
        bz = 2;
      }
  }
  return dim3((uint32_t)bx, (uint32_t)by, (uint32_t)bz);
}

namespace {

dim3 mhd_grid(const Vec3 &ext, const dim3 &block) {
  return dim3((uint32_t)((ext.x + block.x - 1) / block.x),
              (uint32_t)((ext.y + block.y - 1) / block.y),
              (uint32_t)((ext.z + block.z - 1) / block.z));
}

void mhd_div_launch_on(LocalDomain &d, const Rect3 &region, const MhdCoeffs &cf,
                       hipStream_t stream) {
  const Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  MhdParams p{};
  mhd_fill_params(d, region, cf, p);
  dim3 block = mhd_block();
  dim3 grid = mhd_grid(ext, block);
  p.ychunk = mhd_ychunk((int32_t)grid.y);
  if (block.x * block.y * block.z > 256)
    hipLaunchKernelGGL(mhd_div_kernel_b512, grid, block, 0, stream, p);
  else
    hipLaunchKernelGGL(mhd_div_kernel, grid, block, 0, stream, p);
  STENCIL_HIP(hipGetLastError());
}

void mhd_substep_launch_on(LocalDomain &d, const Rect3 &region, int step, double dt,
                           const MhdCoeffs &cf, hipStream_t sScalar, hipStream_t sMomentum) {
  const Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  // Williamson (1980) coefficients
  static const double ALPHA[3] = {0.0, -5.0 / 9.0, -153.0 / 128.0};
  static const double BETA[3] = {1.0 / 3.0, 15.0 / 16.0, 8.0 / 15.0};
  MhdParams p{};
  mhd_fill_params(d, region, cf, p);
  p.dt = dt;
  p.alpha_over_beta_prev = (step == 0) ? 0.0 : ALPHA[step] / BETA[step - 1];
  p.beta = BETA[step];
  dim3 block = mhd_block();
  dim3 grid = mhd_grid(ext, block);
  p.ychunk = mhd_ychunk((int32_t)grid.y);
  // scalar (writes lnrho/ss/aa) and momentum (writes uu) touch disjoint
  // outputs and only read shared inputs: run them CONCURRENTLY on the two
  // streams (the caller joins them)
  const bool b512 = block.x * block.y * block.z > 256;
  if (b512)
    hipLaunchKernelGGL(mhd_scalar_kernel_b512, grid, block, 0, sScalar, p);
  else
    hipLaunchKernelGGL(mhd_scalar_kernel, grid, block, 0, sScalar, p);
  STENCIL_HIP(hipGetLastError());
  static int mom5 = -1, zmarch = -1, mom2 = -1;
  if (mom5 < 0) {
    const char *e = getenv("STENCIL_MHD_MOM5");
    mom5 = (e && e[0] == '1') ? 1 : 0;
    e = getenv("STENCIL_MHD_ZMARCH");
    zmarch = (e && e[0] == '1') ? 1 : 0;
    e = getenv("STENCIL_MHD_MOM2");
    mom2 = (e && e[0] == '1') ? 1 : 0;
  }
  if (zmarch) {
    const dim3 zblock(64, 2, 1);
    const dim3 zgrid((uint32_t)((ext.x + 63) / 64), (uint32_t)((ext.y + 1) / 2), 1);
    hipLaunchKernelGGL(mhd_momentum_zmarch_kernel, zgrid, zblock, 0, sMomentum, p);
  } else if (mom2) {
    const dim3 pgrid(grid.x, grid.y, (uint32_t)((ext.z + 2 * block.z - 1) / (2 * block.z)));
    hipLaunchKernelGGL(mhd_momentum_pair_kernel, pgrid, block, 0, sMomentum, p);
  } else if (b512)
    hipLaunchKernelGGL(mhd_momentum_kernel_b512, grid, block, 0, sMomentum, p);
  else if (mom5)
    hipLaunchKernelGGL(mhd_momentum_kernel_w5, grid, block, 0, sMomentum, p);
  else
    hipLaunchKernelGGL(mhd_momentum_kernel, grid, block, 0, sMomentum, p);
  STENCIL_HIP(hipGetLastError());
}

} // namespace

void mhd_div_pass(ExchangeEngine &eng, int dom, const Rect3 &region, const MhdCoeffs &cf,
                  int streamId) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  mhd_div_launch_on(d, region, cf, eng.compute_stream(dom, streamId));
}

void mhd_substep(ExchangeEngine &eng, int dom, const Rect3 &region, int step, double dt,
                 const MhdCoeffs &cf, int streamId) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  static int momsplit = -1;
  if (momsplit < 0) {
    const char *e = getenv("STENCIL_MHD_MOMSPLIT");
    momsplit = (e && e[0] == '1') ? 1 : 0;
  }
  if (momsplit) {
    // scalar + three per-component momentum kernels on four concurrent
    // streams (see mhd_momentum_comp_kernel)
    const Vec3 ext = region.extent();
    if (ext.flatten() <= 0) return;
    static const double ALPHA[3] = {0.0, -5.0 / 9.0, -153.0 / 128.0};
    static const double BETA[3] = {1.0 / 3.0, 15.0 / 16.0, 8.0 / 15.0};
    MhdParams p{};
    mhd_fill_params(d, region, cf, p);
    p.dt = dt;
    p.alpha_over_beta_prev = (step == 0) ? 0.0 : ALPHA[step] / BETA[step - 1];
    p.beta = BETA[step];
    dim3 block = mhd_block();
    dim3 grid = mhd_grid(ext, block);
    p.ychunk = mhd_ychunk((int32_t)grid.y);
    hipLaunchKernelGGL(mhd_scalar_kernel, grid, block, 0, eng.compute_stream(dom, streamId), p);
    hipLaunchKernelGGL(mhd_momentum_comp_kernel<0>, grid, block, 0, eng.compute_stream(dom, 1),
                       p);
    hipLaunchKernelGGL(mhd_momentum_comp_kernel<1>, grid, block, 0, eng.compute_stream(dom, 2),
                       p);
    hipLaunchKernelGGL(mhd_momentum_comp_kernel<2>, grid, block, 0, eng.compute_stream(dom, 3),
                       p);
    STENCIL_HIP(hipGetLastError());
    return;
  }
  mhd_substep_launch_on(d, region, step, dt, cf, eng.compute_stream(dom, streamId),
                        eng.compute_stream(dom, 1 - streamId));
}

namespace {

// Whole-substep replay graphs for the single-process single-domain MHD
// path (the world=1 astaroth shape). Each RK3 substep s captures, per
// buffer parity: [X1 translates (field halos, group 0) -> div pass ->
// X2 translates (div halos, group 1) -> scalar || momentum (event-forked
// second stream) -> device-side table swap]. The host gap measured
// ~0.45 ms per substep at 256^3 (profiles/astaroth_256_kernel_stats.csv
// GPU-busy vs wall); replay costs one hipGraphLaunch.
struct MhdStepGraph {
  hipStream_t stream = nullptr;  // capture/replay stream
  hipStream_t stream2 = nullptr; // momentum fork
  hipEvent_t evFork = nullptr, evJoin = nullptr;
  hipGraphExec_t exec[3][2] = {{nullptr, nullptr}, {nullptr, nullptr}, {nullptr, nullptr}};
  int parity = 0;
  ExchangeEngine *eng = nullptr;
  int dom = 0;
};
std::vector<std::unique_ptr<MhdStepGraph>> g_mhdGraphs;

} // namespace

int64_t mhd_graph_create(ExchangeEngine &eng, int dom, const Rect3 &region, double dt,
                         const MhdCoeffs &cf) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  auto sg = std::make_unique<MhdStepGraph>();
  sg->eng = &eng;
  sg->dom = dom;
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream, hipStreamNonBlocking));
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream2, hipStreamNonBlocking));
  STENCIL_HIP(hipEventCreateWithFlags(&sg->evFork, hipEventDisableTiming));
  STENCIL_HIP(hipEventCreateWithFlags(&sg->evJoin, hipEventDisableTiming));
  for (int s = 0; s < 3; ++s) {
    for (int par = 0; par < 2; ++par) {
      STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
      eng.launch_translates_plain_on((uintptr_t)sg->stream, 0); // X1: field halos
      mhd_div_launch_on(d, region, cf, sg->stream);
      eng.launch_translates_plain_on((uintptr_t)sg->stream, 1); // X2: div halos
      STENCIL_HIP(hipEventRecord(sg->evFork, sg->stream));
      STENCIL_HIP(hipStreamWaitEvent(sg->stream2, sg->evFork, 0));
      mhd_substep_launch_on(d, region, s, dt, cf, sg->stream, sg->stream2);
      STENCIL_HIP(hipEventRecord(sg->evJoin, sg->stream2));
      STENCIL_HIP(hipStreamWaitEvent(sg->stream, sg->evJoin, 0));
      d.enqueue_table_swap(sg->stream);
      hipGraph_t g = nullptr;
      STENCIL_HIP(hipStreamEndCapture(sg->stream, &g));
      STENCIL_HIP(hipGraphInstantiate(&sg->exec[s][par], g, nullptr, nullptr, 0));
      STENCIL_HIP(hipGraphDestroy(g));
      d.swap(); // bake the other parity next
    }
  }
  g_mhdGraphs.push_back(std::move(sg));
  return (int64_t)g_mhdGraphs.size() - 1;
}

void mhd_graph_iter(int64_t handle, int64_t nIters) {
  MhdStepGraph &sg = *g_mhdGraphs.at(handle);
  LocalDomain &d = sg.eng->domain(sg.dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  for (int64_t i = 0; i < nIters; ++i)
    for (int s = 0; s < 3; ++s) {
      STENCIL_HIP(hipGraphLaunch(sg.exec[s][sg.parity], sg.stream));
      sg.parity ^= 1;
      d.swap_host_only();
    }
}

void mhd_graph_sync(int64_t handle) {
  MhdStepGraph &sg = *g_mhdGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipStreamSynchronize(sg.stream));
}

// Multi-rank substep graphs (one rank, one domain, cross-rank halos via
// IPC -- the 8-GPU single-node weak-scaling shape). Each RK3 substep
// splits at the TWO cross-rank barriers (field halos X1, div halos X2):
//   G1[par]    = [div(interior) || (translates g0 + staged packs g0)]
//   <barrier: every rank's field halos written>
//   G2[s][par] = [staged unpacks g0 -> div(exteriors) ->
//                 (translates g1 + staged packs g1) || substep(interior)]
//   <barrier: every rank's div halos written>
//   G3[s][par] = [staged unpacks g1 -> substep(exteriors) ->
//                 device table swap + view flips]
// The skew-safety argument is the jacobi one (jacobi_mr_graph_*): each
// barrier bounds peers to one phase, staged buffers are double-buffered
// by parity, and direct writes land in halo rings disjoint from locally
// written compute cells.
namespace {
struct MhdMrGraph {
  hipStream_t stream = nullptr, stream2 = nullptr, stream3 = nullptr;
  hipEvent_t evF = nullptr, evJ2 = nullptr, evJ3 = nullptr;
  hipGraphExec_t g1[2] = {nullptr, nullptr};
  hipGraphExec_t g2[3][2] = {};
  hipGraphExec_t g3[3][2] = {};
  int parity = 0, substep = 0;
  ExchangeEngine *eng = nullptr;
  int dom = 0;
};
std::vector<std::unique_ptr<MhdMrGraph>> g_mhdMrGraphs;
} // namespace

int64_t mhd_mr_graph_create(ExchangeEngine &eng, int dom, const Rect3 &interior,
                            const std::vector<Rect3> &exteriors, double dt, const MhdCoeffs &cf) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  auto sg = std::make_unique<MhdMrGraph>();
  sg->eng = &eng;
  sg->dom = dom;
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream, hipStreamNonBlocking));
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream2, hipStreamNonBlocking));
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream3, hipStreamNonBlocking));
  STENCIL_HIP(hipEventCreateWithFlags(&sg->evF, hipEventDisableTiming));
  STENCIL_HIP(hipEventCreateWithFlags(&sg->evJ2, hipEventDisableTiming));
  STENCIL_HIP(hipEventCreateWithFlags(&sg->evJ3, hipEventDisableTiming));
  auto instantiate = [&](hipGraphExec_t &exec) {
    hipGraph_t g = nullptr;
    STENCIL_HIP(hipStreamEndCapture(sg->stream, &g));
    STENCIL_HIP(hipGraphInstantiate(&exec, g, nullptr, nullptr, 0));
    STENCIL_HIP(hipGraphDestroy(g));
  };
  for (int par = 0; par < 2; ++par) {
    // G1: div(interior) concurrent with the outgoing field halos
    STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
    STENCIL_HIP(hipEventRecord(sg->evF, sg->stream));
    STENCIL_HIP(hipStreamWaitEvent(sg->stream2, sg->evF, 0));
    mhd_div_launch_on(d, interior, cf, sg->stream);
    eng.launch_translates_plain_on((uintptr_t)sg->stream2, 0);
    eng.launch_packs_plain_on((uintptr_t)sg->stream2, 1 + par);
    STENCIL_HIP(hipEventRecord(sg->evJ2, sg->stream2));
    STENCIL_HIP(hipStreamWaitEvent(sg->stream, sg->evJ2, 0));
    instantiate(sg->g1[par]);
    for (int s = 0; s < 3; ++s) {
      // G2: incoming field halos -> div(ext) -> div halos out || interior
      STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
      eng.launch_unpacks_plain_on((uintptr_t)sg->stream, 1 + par);
      for (const Rect3 &box : exteriors) mhd_div_launch_on(d, box, cf, sg->stream);
      STENCIL_HIP(hipEventRecord(sg->evF, sg->stream));
      STENCIL_HIP(hipStreamWaitEvent(sg->stream2, sg->evF, 0));
      STENCIL_HIP(hipStreamWaitEvent(sg->stream3, sg->evF, 0));
      eng.launch_translates_plain_on((uintptr_t)sg->stream, 1);
      eng.launch_packs_plain_on((uintptr_t)sg->stream, 4 + par); // 3*1+1+par
      mhd_substep_launch_on(d, interior, s, dt, cf, sg->stream2, sg->stream3);
      STENCIL_HIP(hipEventRecord(sg->evJ2, sg->stream2));
      STENCIL_HIP(hipEventRecord(sg->evJ3, sg->stream3));
      STENCIL_HIP(hipStreamWaitEvent(sg->stream, sg->evJ2, 0));
      STENCIL_HIP(hipStreamWaitEvent(sg->stream, sg->evJ3, 0));
      instantiate(sg->g2[s][par]);
      // G3: incoming div halos -> exterior shells -> swap
      STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
      eng.launch_unpacks_plain_on((uintptr_t)sg->stream, 4 + par);
      for (const Rect3 &box : exteriors)
        mhd_substep_launch_on(d, box, s, dt, cf, sg->stream, sg->stream);
      d.enqueue_table_swap(sg->stream);
      eng.enqueue_view_flips((uintptr_t)sg->stream);
      instantiate(sg->g3[s][par]);
    }
    d.swap(); // bake the other parity's kernarg pointers next round
  }
  g_mhdMrGraphs.push_back(std::move(sg));
  return (int64_t)g_mhdMrGraphs.size() - 1;
}

uintptr_t mhd_mr_graph_stream(int64_t handle) {
  return (uintptr_t)g_mhdMrGraphs.at(handle)->stream;
}

void mhd_mr_phase1(int64_t handle) {
  MhdMrGraph &sg = *g_mhdMrGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipGraphLaunch(sg.g1[sg.parity], sg.stream));
}

void mhd_mr_phase2(int64_t handle) {
  MhdMrGraph &sg = *g_mhdMrGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipGraphLaunch(sg.g2[sg.substep][sg.parity], sg.stream));
}

void mhd_mr_phase3(int64_t handle) {
  MhdMrGraph &sg = *g_mhdMrGraphs.at(handle);
  LocalDomain &d = sg.eng->domain(sg.dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  STENCIL_HIP(hipGraphLaunch(sg.g3[sg.substep][sg.parity], sg.stream));
  sg.substep = (sg.substep + 1) % 3;
  sg.parity ^= 1;
  d.swap_host_only(); // in-graph kernels flip the device state
  sg.eng->flip_views_host_only();
}

void mhd_mr_graph_sync(int64_t handle) {
  MhdMrGraph &sg = *g_mhdMrGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipStreamSynchronize(sg.stream));
}

} // namespace stencil_amd
