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
  const char *curr[8];
  char *next[8];
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
};

enum { LNRHO = 0, UUX = 1, UUY = 2, UUZ = 3, AAX = 4, AAY = 5, AAZ = 6, SS = 7 };

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
  const char *base[8]; // per-field pointer AT the cell
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
  const char *base[8];
  int64_t pitch, plane;
  char *out[8];
};

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
  for (int q = 0; q < 8; ++q) {
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
__global__ void __launch_bounds__(256) mhd_scalar_kernel(MhdParams p) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz = blockIdx.z * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 8; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const Vec3d uu = {st.c(UUX), st.c(UUY), st.c(UUZ)};
  {
    const Vec3d glnrho = st.grad(LNRHO, ix, iy, iz);
    const double divu = st.dx(UUX, ix) + st.dy(UUY, iy) + st.dz(UUZ, iz);
    write_rk3(p, st, c.out[LNRHO], LNRHO, -dot(uu, glnrho) - divu);
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

// kernel 2a: Lorentz force j x B into a scratch array (j needs the cross
// derivatives of A -- the most load-heavy part of the solver; isolating it
// keeps each kernel's register set small enough for >=2 waves/SIMD)
struct MhdScratch {
  char *ptr;           // 3 consecutive (z,y,x) fp64 arrays over the region
  int64_t rowStride;   // extX * 8
  int64_t planeStride; // rowStride * extY
  int64_t compStride;  // planeStride * extZ
};

__global__ void __launch_bounds__(256) mhd_lorentz_kernel(MhdParams p, MhdScratch sc) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz = blockIdx.z * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 8; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  char *out = sc.ptr + (int64_t)lz * sc.planeStride + (int64_t)ly * sc.rowStride + (int64_t)lx * 8;
  // j = grad(div A) - lap(A), expanded per component so each is a small
  // independent expression (2 cross + 2 second derivatives) and the
  // register live-set stays flat:
  //   j.x = dxy Ay + dxz Az - dyy Ax - dzz Ax   (etc. cyclically)
  {
    const double jx = st.dxy(AAY, ix, iy) + st.dxz(AAZ, ix, iz) - st.dyy(AAX, iy * iy) -
                      st.dzz(AAX, iz * iz);
    __builtin_nontemporal_store(jx, (double *)out);
  }
  {
    const double jy = st.dxy(AAX, ix, iy) + st.dyz(AAZ, iy, iz) - st.dxx(AAY, ix * ix) -
                      st.dzz(AAY, iz * iz);
    __builtin_nontemporal_store(jy, (double *)(out + sc.compStride));
  }
  {
    const double jz = st.dxz(AAX, ix, iz) + st.dyz(AAY, iy, iz) - st.dxx(AAZ, ix * ix) -
                      st.dyy(AAZ, iy * iy);
    __builtin_nontemporal_store(jz, (double *)(out + 2 * sc.compStride));
  }
}

// kernel 2b: momentum update (advection + pressure + viscosity + the
// precomputed Lorentz force)
__global__ void __launch_bounds__(256, 3) mhd_momentum_kernel(MhdParams p, MhdScratch sc) {
  const int32_t lx = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz = blockIdx.z * blockDim.z + threadIdx.z;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;
  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 8; ++q) st.base[q] = c.base[q];
  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const Vec3d uu = {st.c(UUX), st.c(UUY), st.c(UUZ)};
  const double rho_inv = exp(-st.c(LNRHO));
  const char *jb =
      sc.ptr + (int64_t)lz * sc.planeStride + (int64_t)ly * sc.rowStride + (int64_t)lx * 8;
  const Vec3d B = {st.dy(AAZ, 1.0 / p.dsy) - st.dz(AAY, 1.0 / p.dsz),
                   st.dz(AAX, 1.0 / p.dsz) - st.dx(AAZ, 1.0 / p.dsx),
                   st.dx(AAY, 1.0 / p.dsx) - st.dy(AAX, 1.0 / p.dsy)};
  const Vec3d j = {*(const double *)jb, *(const double *)(jb + sc.compStride),
                   *(const double *)(jb + 2 * sc.compStride)};
  const Vec3d jxB = cross(j, B);

  {
    const double ugradu = uu.x * st.dx(UUX, ix) + uu.y * st.dy(UUX, iy) + uu.z * st.dz(UUX, iz);
    const double press = st.dx(LNRHO, ix) + p.cp_inv * st.dx(SS, ix);
    const double graddiv = st.dxx(UUX, ix * ix) + st.dxy(UUY, ix, iy) + st.dxz(UUZ, ix, iz);
    const double visc = p.nu * (st.lap(UUX, ix, iy, iz) + graddiv / 3.0);
    write_rk3(p, st, c.out[UUX], UUX, -ugradu - p.cs2 * press + rho_inv * jxB.x + visc);
  }
  {
    const double ugradu = uu.x * st.dx(UUY, ix) + uu.y * st.dy(UUY, iy) + uu.z * st.dz(UUY, iz);
    const double press = st.dy(LNRHO, iy) + p.cp_inv * st.dy(SS, iy);
    const double graddiv = st.dxy(UUX, ix, iy) + st.dyy(UUY, iy * iy) + st.dyz(UUZ, iy, iz);
    const double visc = p.nu * (st.lap(UUY, ix, iy, iz) + graddiv / 3.0);
    write_rk3(p, st, c.out[UUY], UUY, -ugradu - p.cs2 * press + rho_inv * jxB.y + visc);
  }
  {
    const double ugradu = uu.x * st.dx(UUZ, ix) + uu.y * st.dy(UUZ, iy) + uu.z * st.dz(UUZ, iz);
    const double press = st.dz(LNRHO, iz) + p.cp_inv * st.dz(SS, iz);
    const double graddiv = st.dxz(UUX, ix, iz) + st.dyz(UUY, iy, iz) + st.dzz(UUZ, iz * iz);
    const double visc = p.nu * (st.lap(UUZ, ix, iy, iz) + graddiv / 3.0);
    write_rk3(p, st, c.out[UUZ], UUZ, -ugradu - p.cs2 * press + rho_inv * jxB.z + visc);
  }
}

// LDS-tiled momentum variant (STENCIL_MHD_LDS=1): stages the 6 vector
// fields (uu, aa) of a (32,4,2)-cell tile + radius-3 halo in LDS
// (145.9 KB -> one block per CU) and evaluates j, B, the Lorentz force,
// advection, viscosity and grad(div u) from LDS lines, replacing BOTH the
// Lorentz and momentum global-line kernels. lnrho/ss pressure gradients
// (12 loads/cell) stay on the global path.
#define MT_X 32
#define MT_Y 4
#define MT_Z 2
#define MH 3 // halo

struct LdsTile {
  // [field][z][y][x]; +6 halo cells per axis
  double t[6][MT_Z + 6][MT_Y + 6][MT_X + 6];
};

__global__ void __launch_bounds__(256) mhd_momentum_lds_kernel(MhdParams p) {
  __shared__ LdsTile s;
  const int32_t gx0 = blockIdx.x * MT_X;
  const int32_t gy0 = blockIdx.y * MT_Y;
  const int32_t gz0 = blockIdx.z * MT_Z;

  // cooperative load (fields 0..5 = UUX..AAZ, q = 1 + f)
  const int64_t rawX = p.extX; // region extents (guards below)
  constexpr int TILE = 6 * (MT_Z + 6) * (MT_Y + 6) * (MT_X + 6);
  for (int i = threadIdx.x; i < TILE; i += 256) {
    int r = i;
    const int x = r % (MT_X + 6);
    r /= (MT_X + 6);
    const int y = r % (MT_Y + 6);
    r /= (MT_Y + 6);
    const int z = r % (MT_Z + 6);
    const int f = r / (MT_Z + 6);
    // global coords of this tile cell (region-local then absolute)
    const int64_t lx = (int64_t)gx0 + x - MH;
    const int64_t ly = (int64_t)gy0 + y - MH;
    const int64_t lz = (int64_t)gz0 + z - MH;
    // clamp into the full allocation (out-of-range cells are never read
    // by an in-region output; clamping just keeps the address legal)
    const int64_t axm = p.loX + lx - p.allocX;
    const int64_t aym = p.loY + ly - p.allocY;
    const int64_t azm = p.loZ + lz - p.allocZ;
    const int64_t ax = axm < 0 ? 0 : axm;
    const int64_t ay = aym < 0 ? 0 : aym;
    const int64_t az = azm < 0 ? 0 : azm;
    const char *base = p.curr[1 + f];
    s.t[f][z][y][x] = *(const double *)(base + az * p.plane + ay * p.pitch + ax * 8);
  }
  __syncthreads();

  const int32_t cx = threadIdx.x % MT_X;
  const int32_t cy = (threadIdx.x / MT_X) % MT_Y;
  const int32_t cz = threadIdx.x / (MT_X * MT_Y);
  const int32_t lx = gx0 + cx, ly = gy0 + cy, lz = gz0 + cz;
  if (lx >= p.extX || ly >= p.extY || lz >= p.extZ) return;

  const double ix = 1.0 / p.dsx, iy = 1.0 / p.dsy, iz = 1.0 / p.dsz;
  const int X = cx + MH, Y = cy + MH, Z = cz + MH;
  enum { FUX = 0, FUY = 1, FUZ = 2, FAX = 3, FAY = 4, FAZ = 5 };

#define LV(f, dxo, dyo, dzo) s.t[f][Z + (dzo)][Y + (dyo)][X + (dxo)]
#define LDX(f) ((D1[0] * (LV(f, 1, 0, 0) - LV(f, -1, 0, 0)) + D1[1] * (LV(f, 2, 0, 0) - LV(f, -2, 0, 0)) + D1[2] * (LV(f, 3, 0, 0) - LV(f, -3, 0, 0))) * ix)
#define LDY(f) ((D1[0] * (LV(f, 0, 1, 0) - LV(f, 0, -1, 0)) + D1[1] * (LV(f, 0, 2, 0) - LV(f, 0, -2, 0)) + D1[2] * (LV(f, 0, 3, 0) - LV(f, 0, -3, 0))) * iy)
#define LDZ(f) ((D1[0] * (LV(f, 0, 0, 1) - LV(f, 0, 0, -1)) + D1[1] * (LV(f, 0, 0, 2) - LV(f, 0, 0, -2)) + D1[2] * (LV(f, 0, 0, 3) - LV(f, 0, 0, -3))) * iz)
#define LDXX(f) ((D2[0] * LV(f, 0, 0, 0) + D2[1] * (LV(f, 1, 0, 0) + LV(f, -1, 0, 0)) + D2[2] * (LV(f, 2, 0, 0) + LV(f, -2, 0, 0)) + D2[3] * (LV(f, 3, 0, 0) + LV(f, -3, 0, 0))) * (ix * ix))
#define LDYY(f) ((D2[0] * LV(f, 0, 0, 0) + D2[1] * (LV(f, 0, 1, 0) + LV(f, 0, -1, 0)) + D2[2] * (LV(f, 0, 2, 0) + LV(f, 0, -2, 0)) + D2[3] * (LV(f, 0, 3, 0) + LV(f, 0, -3, 0))) * (iy * iy))
#define LDZZ(f) ((D2[0] * LV(f, 0, 0, 0) + D2[1] * (LV(f, 0, 0, 1) + LV(f, 0, 0, -1)) + D2[2] * (LV(f, 0, 0, 2) + LV(f, 0, 0, -2)) + D2[3] * (LV(f, 0, 0, 3) + LV(f, 0, 0, -3))) * (iz * iz))

  auto lcross = [&](int f, int a1, int a2, double i1, double i2) {
    // composed first-derivative quadrant sum along axes a1, a2 (0=x,1=y,2=z)
    double acc = 0;
#pragma unroll
    for (int i = 1; i <= 3; ++i)
#pragma unroll
      for (int k = 1; k <= 3; ++k) {
        const int dx1 = (a1 == 0) ? i : 0, dy1 = (a1 == 1) ? i : 0, dz1 = (a1 == 2) ? i : 0;
        const int dx2 = (a2 == 0) ? k : 0, dy2 = (a2 == 1) ? k : 0, dz2 = (a2 == 2) ? k : 0;
        acc += D1[i - 1] * D1[k - 1] *
               (s.t[f][Z + dz1 + dz2][Y + dy1 + dy2][X + dx1 + dx2] -
                s.t[f][Z + dz1 - dz2][Y + dy1 - dy2][X + dx1 - dx2] -
                s.t[f][Z - dz1 + dz2][Y - dy1 + dy2][X - dx1 + dx2] +
                s.t[f][Z - dz1 - dz2][Y - dy1 - dy2][X - dx1 - dx2]);
      }
    return acc * i1 * i2;
  };

  const MhdCommon c = mhd_setup(p, lx, ly, lz);
  Stencil st;
  st.pitch = c.pitch;
  st.plane = c.plane;
#pragma unroll
  for (int q = 0; q < 8; ++q) st.base[q] = c.base[q];

  const Vec3d uu = {LV(FUX, 0, 0, 0), LV(FUY, 0, 0, 0), LV(FUZ, 0, 0, 0)};
  const double rho_inv = exp(-st.c(LNRHO));

  // current and magnetic field from the A tile
  const double jx = lcross(FAY, 0, 1, ix, iy) + lcross(FAZ, 0, 2, ix, iz) - LDYY(FAX) - LDZZ(FAX);
  const double jy = lcross(FAX, 0, 1, ix, iy) + lcross(FAZ, 1, 2, iy, iz) - LDXX(FAY) - LDZZ(FAY);
  const double jz = lcross(FAX, 0, 2, ix, iz) + lcross(FAY, 1, 2, iy, iz) - LDXX(FAZ) - LDYY(FAZ);
  const Vec3d B = {LDY(FAZ) - LDZ(FAY), LDZ(FAX) - LDX(FAZ), LDX(FAY) - LDY(FAX)};
  const Vec3d jxB = cross({jx, jy, jz}, B);

  {
    const double ugradu = uu.x * LDX(FUX) + uu.y * LDY(FUX) + uu.z * LDZ(FUX);
    const double press = st.dx(LNRHO, ix) + p.cp_inv * st.dx(SS, ix);
    const double graddiv = LDXX(FUX) + lcross(FUY, 0, 1, ix, iy) + lcross(FUZ, 0, 2, ix, iz);
    const double visc = p.nu * (LDXX(FUX) + LDYY(FUX) + LDZZ(FUX) + graddiv / 3.0);
    write_rk3(p, st, c.out[UUX], UUX, -ugradu - p.cs2 * press + rho_inv * jxB.x + visc);
  }
  {
    const double ugradu = uu.x * LDX(FUY) + uu.y * LDY(FUY) + uu.z * LDZ(FUY);
    const double press = st.dy(LNRHO, iy) + p.cp_inv * st.dy(SS, iy);
    const double graddiv = lcross(FUX, 0, 1, ix, iy) + LDYY(FUY) + lcross(FUZ, 1, 2, iy, iz);
    const double visc = p.nu * (LDXX(FUY) + LDYY(FUY) + LDZZ(FUY) + graddiv / 3.0);
    write_rk3(p, st, c.out[UUY], UUY, -ugradu - p.cs2 * press + rho_inv * jxB.y + visc);
  }
  {
    const double ugradu = uu.x * LDX(FUZ) + uu.y * LDY(FUZ) + uu.z * LDZ(FUZ);
    const double press = st.dz(LNRHO, iz) + p.cp_inv * st.dz(SS, iz);
    const double graddiv = lcross(FUX, 0, 2, ix, iz) + lcross(FUY, 1, 2, iy, iz) + LDZZ(FUZ);
    const double visc = p.nu * (LDXX(FUZ) + LDYY(FUZ) + LDZZ(FUZ) + graddiv / 3.0);
    write_rk3(p, st, c.out[UUZ], UUZ, -ugradu - p.cs2 * press + rho_inv * jxB.z + visc);
  }
}
#undef LV
#undef LDX
#undef LDY
#undef LDZ
#undef LDXX
#undef LDYY
#undef LDZZ

} // namespace

void mhd_substep(ExchangeEngine &eng, int dom, const Rect3 &region, int step, double dt,
                 const MhdCoeffs &cf, int64_t scratchBuf, int streamId) {
  LocalDomain &d = eng.domain(dom);
  if (d.num_data() != 8 || d.elem_size(0) != 8)
    throw std::runtime_error("mhd_substep: domain must have 8 fp64 quantities");
  const Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  // Williamson (1980) coefficients
  static const double ALPHA[3] = {0.0, -5.0 / 9.0, -153.0 / 128.0};
  static const double BETA[3] = {1.0 / 3.0, 15.0 / 16.0, 8.0 / 15.0};
  MhdParams p{};
  for (int q = 0; q < 8; ++q) {
    p.curr[q] = d.curr(q).ptr;
    p.next[q] = d.next(q).ptr;
  }
  p.pitch = d.curr(0).pitch;
  p.plane = d.curr(0).plane();
  for (int q = 1; q < 8; ++q)
    if (d.curr(q).pitch != p.pitch) throw std::runtime_error("mhd: field pitches differ");
  const Rect3 full = d.full_region();
  p.allocX = full.lo.x;
  p.allocY = full.lo.y;
  p.allocZ = full.lo.z;
  p.loX = region.lo.x;
  p.loY = region.lo.y;
  p.loZ = region.lo.z;
  p.extX = (int32_t)ext.x;
  p.extY = (int32_t)ext.y;
  p.extZ = (int32_t)ext.z;
  p.dsx = cf.dsx;
  p.dsy = cf.dsy;
  p.dsz = cf.dsz;
  p.dt = dt;
  p.cs2 = cf.cs2;
  p.cp_inv = cf.cp_inv;
  p.nu = cf.nu;
  p.eta = cf.eta;
  p.chi = cf.chi;
  p.alpha_over_beta_prev = (step == 0) ? 0.0 : ALPHA[step] / BETA[step - 1];
  p.beta = BETA[step];
  STENCIL_HIP(hipSetDevice(d.gpu()));
  // 32x4x2 measured best on gfx950 (block sweep in gpurun_out/gpu6.log:
  // 3D-ish tiles reuse the y/z derivative lines within a block)
  static int bx = 0, by = 0, bz = 0;
  if (!bx) {
    bx = 32;
    by = 4;
    bz = 2;
    if (const char *e = getenv("STENCIL_MHD_BLOCK"))
      if (sscanf(e, "%dx%dx%d", &bx, &by, &bz) != 3 || bx * by * bz != 256) {
        bx = 32;
        by = 4;
        bz = 2;
      }
  }
  dim3 block((uint32_t)bx, (uint32_t)by, (uint32_t)bz);
  dim3 grid((uint32_t)((ext.x + bx - 1) / bx), (uint32_t)((ext.y + by - 1) / by),
            (uint32_t)((ext.z + bz - 1) / bz));
  static int useLds = -1;
  if (useLds < 0) {
    const char *e = getenv("STENCIL_MHD_LDS");
    useLds = (e && e[0] == '1') ? 1 : 0;
  }
  if (useLds) {
    hipLaunchKernelGGL(mhd_scalar_kernel, grid, block, 0, eng.compute_stream(dom, streamId), p);
    STENCIL_HIP(hipGetLastError());
    dim3 lblock(256, 1, 1);
    dim3 lgrid((uint32_t)((ext.x + 31) / 32), (uint32_t)((ext.y + 3) / 4),
               (uint32_t)((ext.z + 1) / 2));
    hipLaunchKernelGGL(mhd_momentum_lds_kernel, lgrid, lblock, 0, eng.compute_stream(dom, streamId),
                       p);
    STENCIL_HIP(hipGetLastError());
    return;
  }
  MhdScratch sc{};
  sc.rowStride = ext.x * 8;
  sc.planeStride = sc.rowStride * ext.y;
  sc.compStride = sc.planeStride * ext.z;
  if (eng.buffer_bytes(scratchBuf) < 3 * sc.compStride)
    throw std::runtime_error("mhd_substep: scratch buffer too small for region");
  sc.ptr = (char *)eng.buffer_ptr(scratchBuf);
  hipLaunchKernelGGL(mhd_scalar_kernel, grid, block, 0, eng.compute_stream(dom, streamId), p);
  STENCIL_HIP(hipGetLastError());
  hipLaunchKernelGGL(mhd_lorentz_kernel, grid, block, 0, eng.compute_stream(dom, streamId), p, sc);
  STENCIL_HIP(hipGetLastError());
  hipLaunchKernelGGL(mhd_momentum_kernel, grid, block, 0, eng.compute_stream(dom, streamId), p, sc);
  STENCIL_HIP(hipGetLastError());
}

} // namespace stencil_amd
