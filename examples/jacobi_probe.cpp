// Jacobi interior-kernel ablation probe (within-probe interleaved A/B):
// isolates what limits the 7-point kernel on gfx950. Variants share the
// thread mapping of jacobi_kernel_v4 (4-wide float4 x-strips, 16-z march).
//   copy     : out = center (pure stream; the roofline for this mapping)
//   xz       : x + rolled z neighbors only (no y-row loads)
//   full     : the real 7-point stencil
//   full8    : 8 cells per thread (2 float4 strips)
// Usage: jacobi_probe [n=752] [rounds=10]
#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <hip/hip_runtime.h>

#define CHECK(x)                                                                                   \
  if ((x) != hipSuccess) {                                                                         \
    printf("hip error %s @%d\n", hipGetErrorString(hipGetLastError()), __LINE__);                  \
    exit(1);                                                                                       \
  }

typedef float vfloat4 __attribute__((ext_vector_type(4)));

struct P {
  const char *src;
  char *dst;
  int64_t pitch, plane;
  int32_t nx4, ny, nz; // nx4 = float4 units per row
};

#define ZCH 16

template <int VARIANT>
__global__ void __launch_bounds__(256) probe(P p) {
  const int32_t u = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t y = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t z0 = blockIdx.z * ZCH;
  if (u >= p.nx4 || y >= p.ny) return;
  const int32_t zEnd = min(z0 + ZCH, p.nz);
  const char *col = p.src + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  char *dcol = p.dst + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  float4 cm = *(const float4 *)(col - p.plane);
  float4 cc = *(const float4 *)(col);
  for (int32_t z = z0; z < zEnd; ++z) {
    float4 out;
    if (VARIANT == 0) { // pure copy
      out = cc;
    } else {
      const float4 cp = *(const float4 *)(col + p.plane);
      const float left = *(const float *)(col - 4);
      const float right = *(const float *)(col + 16);
      if (VARIANT == 1) { // x + z only
        out.x = (cc.y + left + cp.x + cm.x) / 6.0f;
        out.y = (cc.z + cc.x + cp.y + cm.y) / 6.0f;
        out.z = (cc.w + cc.y + cp.z + cm.z) / 6.0f;
        out.w = (right + cc.z + cp.w + cm.w) / 6.0f;
      } else { // full
        const float4 py = *(const float4 *)(col + p.pitch);
        const float4 my = *(const float4 *)(col - p.pitch);
        out.x = (cc.y + left + py.x + my.x + cp.x + cm.x) / 6.0f;
        out.y = (cc.z + cc.x + py.y + my.y + cp.y + cm.y) / 6.0f;
        out.z = (cc.w + cc.y + py.z + my.z + cp.z + cm.z) / 6.0f;
        out.w = (right + cc.z + py.w + my.w + cp.w + cm.w) / 6.0f;
      }
      cm = cc;
      cc = cp;
    }
    if (VARIANT == 3) { // full stencil, PLAIN store (NT ablation)
      *(float4 *)dcol = out;
    } else {
      vfloat4 ov = {out.x, out.y, out.z, out.w};
      __builtin_nontemporal_store(ov, (vfloat4 *)dcol);
    }
    if (VARIANT == 0) {
      cc = *(const float4 *)(col + p.plane);
    }
    col += p.plane;
    dcol += p.plane;
  }
}

// y-neighbor rows staged through LDS: a 64x4 block marching z keeps the
// current plane's center vectors of its 4 rows + 2 halo rows in LDS
// (ping-pong buffers, one barrier per z-step), so the py/my global vector
// loads become LDS reads -- halves the vector-load issue rate.
__global__ void __launch_bounds__(256) probe_lds(P p) {
  __shared__ float4 tile[2][6][64];
  const int32_t tx = threadIdx.x; // x-unit lane
  const int32_t ry = threadIdx.y; // row within block (0..3)
  // no early returns: __syncthreads needs every thread, so out-of-range
  // lanes clamp their coordinates and skip only the store
  const int32_t u0 = blockIdx.x * blockDim.x + tx;
  const int32_t y0 = blockIdx.y * 4 + ry;
  const bool valid = u0 < p.nx4 && y0 < p.ny;
  const int32_t u = min(u0, p.nx4 - 1);
  const int32_t y = min(y0, p.ny - 1);
  const int32_t z0 = blockIdx.z * ZCH;
  const int32_t zEnd = min(z0 + ZCH, p.nz);
  const char *col = p.src + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  char *dcol = p.dst + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  float4 cm = *(const float4 *)(col - p.plane);
  float4 cc = *(const float4 *)(col);
  // prime LDS with plane z0's centers (+2 y-halo rows)
  tile[0][ry + 1][tx] = cc;
  if (ry == 0) tile[0][0][tx] = *(const float4 *)(col - p.pitch);
  if (ry == 3) tile[0][5][tx] = *(const float4 *)(col + p.pitch);
  __syncthreads();
  int buf = 0;
  for (int32_t z = z0; z < zEnd; ++z) {
    const float4 cp = *(const float4 *)(col + p.plane);
    const float left = *(const float *)(col - 4);
    const float right = *(const float *)(col + 16);
    const float4 py = tile[buf][ry + 2][tx];
    const float4 my = tile[buf][ry][tx];
    float4 out;
    out.x = (cc.y + left + py.x + my.x + cp.x + cm.x) / 6.0f;
    out.y = (cc.z + cc.x + py.y + my.y + cp.y + cm.y) / 6.0f;
    out.z = (cc.w + cc.y + py.z + my.z + cp.z + cm.z) / 6.0f;
    out.w = (right + cc.z + py.w + my.w + cp.w + cm.w) / 6.0f;
    if (valid) {
      vfloat4 ov = {out.x, out.y, out.z, out.w};
      __builtin_nontemporal_store(ov, (vfloat4 *)dcol);
    }
    // stage plane z+1 into the other LDS buffer for the next step
    tile[buf ^ 1][ry + 1][tx] = cp;
    if (ry == 0) tile[buf ^ 1][0][tx] = *(const float4 *)(col + p.plane - p.pitch);
    if (ry == 3) tile[buf ^ 1][5][tx] = *(const float4 *)(col + p.plane + p.pitch);
    __syncthreads();
    buf ^= 1;
    cm = cc;
    cc = cp;
    col += p.plane;
    dcol += p.plane;
  }
}

// absolute bandwidth ceiling: flat linear float4 stream over the same bytes
__global__ void __launch_bounds__(256) probe_lin(const char *src, char *dst, int64_t n16) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < n16; i += stride) {
    vfloat4 v = *(const vfloat4 *)(src + i * 16);
    __builtin_nontemporal_store(v, (vfloat4 *)(dst + i * 16));
  }
}

// full stencil with the DISPATCH ORDER swapped: z-chunks advance fastest
// across blockIdx.y, y-rows across blockIdx.z (does walk order engage
// DRAM channels differently?)
__global__ void __launch_bounds__(256) probe_swap(P p) {
  const int32_t u = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t y = blockIdx.z * blockDim.y + threadIdx.y;
  const int32_t z0 = blockIdx.y * ZCH;
  if (u >= p.nx4 || y >= p.ny) return;
  const int32_t zEnd = min(z0 + ZCH, p.nz);
  const char *col = p.src + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  char *dcol = p.dst + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  float4 cm = *(const float4 *)(col - p.plane);
  float4 cc = *(const float4 *)(col);
  for (int32_t z = z0; z < zEnd; ++z) {
    const float4 cp = *(const float4 *)(col + p.plane);
    const float left = *(const float *)(col - 4);
    const float right = *(const float *)(col + 16);
    const float4 py = *(const float4 *)(col + p.pitch);
    const float4 my = *(const float4 *)(col - p.pitch);
    float4 out;
    out.x = (cc.y + left + py.x + my.x + cp.x + cm.x) / 6.0f;
    out.y = (cc.z + cc.x + py.y + my.y + cp.y + cm.y) / 6.0f;
    out.z = (cc.w + cc.y + py.z + my.z + cp.z + cm.z) / 6.0f;
    out.w = (right + cc.z + py.w + my.w + cp.w + cm.w) / 6.0f;
    vfloat4 ov = {out.x, out.y, out.z, out.w};
    __builtin_nontemporal_store(ov, (vfloat4 *)dcol);
    cm = cc;
    cc = cp;
    col += p.plane;
    dcol += p.plane;
  }
}

// 8 cells per thread: two adjacent float4 strips, shared y-row vectors
__global__ void __launch_bounds__(256) probe8(P p) {
  const int32_t u = (blockIdx.x * blockDim.x + threadIdx.x) * 2;
  const int32_t y = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t z0 = blockIdx.z * ZCH;
  if (u + 1 >= p.nx4 || y >= p.ny) return;
  const int32_t zEnd = min(z0 + ZCH, p.nz);
  const char *col = p.src + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  char *dcol = p.dst + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
  float4 cmA = *(const float4 *)(col - p.plane), cmB = *(const float4 *)(col - p.plane + 16);
  float4 ccA = *(const float4 *)(col), ccB = *(const float4 *)(col + 16);
  for (int32_t z = z0; z < zEnd; ++z) {
    const float4 cpA = *(const float4 *)(col + p.plane);
    const float4 cpB = *(const float4 *)(col + p.plane + 16);
    const float left = *(const float *)(col - 4);
    const float right = *(const float *)(col + 32);
    const float4 pyA = *(const float4 *)(col + p.pitch), pyB = *(const float4 *)(col + p.pitch + 16);
    const float4 myA = *(const float4 *)(col - p.pitch), myB = *(const float4 *)(col - p.pitch + 16);
    float4 oA, oB;
    oA.x = (ccA.y + left + pyA.x + myA.x + cpA.x + cmA.x) / 6.0f;
    oA.y = (ccA.z + ccA.x + pyA.y + myA.y + cpA.y + cmA.y) / 6.0f;
    oA.z = (ccA.w + ccA.y + pyA.z + myA.z + cpA.z + cmA.z) / 6.0f;
    oA.w = (ccB.x + ccA.z + pyA.w + myA.w + cpA.w + cmA.w) / 6.0f;
    oB.x = (ccB.y + ccA.w + pyB.x + myB.x + cpB.x + cmB.x) / 6.0f;
    oB.y = (ccB.z + ccB.x + pyB.y + myB.y + cpB.y + cmB.y) / 6.0f;
    oB.z = (ccB.w + ccB.y + pyB.z + myB.z + cpB.z + cmB.z) / 6.0f;
    oB.w = (right + ccB.z + pyB.w + myB.w + cpB.w + cmB.w) / 6.0f;
    vfloat4 a = {oA.x, oA.y, oA.z, oA.w}, b = {oB.x, oB.y, oB.z, oB.w};
    __builtin_nontemporal_store(a, (vfloat4 *)dcol);
    __builtin_nontemporal_store(b, (vfloat4 *)(dcol + 16));
    cmA = ccA;
    cmB = ccB;
    ccA = cpA;
    ccB = cpB;
    col += p.plane;
    dcol += p.plane;
  }
}

// variant mirroring the PRODUCTION kernel structure: slot indirection for
// the base pointers, per-z sphere bounding test, head/tail scalar units
__global__ void __launch_bounds__(256) probe_prod(P p, const char *const *srcSlot,
                                                  char *const *dstSlot, int32_t withSphere) {
  const int32_t u = blockIdx.x * blockDim.x + threadIdx.x;
  const int32_t y = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t z0 = blockIdx.z * ZCH;
  if (y >= p.ny) return;
  const char *srcBase = *srcSlot;
  char *dstBase = *dstSlot;
  const int32_t zEnd = min(z0 + ZCH, p.nz);
  const int32_t body4 = p.nx4;
  if (u < body4) {
    const char *col = srcBase + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
    char *dcol = dstBase + (int64_t)(z0 + 3) * p.plane + (int64_t)(y + 3) * p.pitch + 16 + u * 16;
    float4 cm = *(const float4 *)(col - p.plane);
    float4 cc = *(const float4 *)(col);
    for (int32_t z = z0; z < zEnd; ++z) {
      const float4 cp = *(const float4 *)(col + p.plane);
      const float left = *(const float *)(col - 4);
      const float right = *(const float *)(col + 16);
      const float4 py = *(const float4 *)(col + p.pitch);
      const float4 my = *(const float4 *)(col - p.pitch);
      float4 out;
      out.x = (cc.y + left + py.x + my.x + cp.x + cm.x) / 6.0f;
      out.y = (cc.z + cc.x + py.y + my.y + cp.y + cm.y) / 6.0f;
      out.z = (cc.w + cc.y + py.z + my.z + cp.z + cm.z) / 6.0f;
      out.w = (right + cc.z + py.w + my.w + cp.w + cm.w) / 6.0f;
      if (withSphere & 1) {
        const int32_t dy = y - p.ny / 2, dz = z - p.nz / 2;
        const int32_t yz2 = dy * dy + dz * dz;
        const int32_t srad = p.nx4 * 4 / 10;
        if (yz2 < (srad + 1) * (srad + 1)) {
#pragma unroll
          for (int i = 0; i < 4; ++i) {
            const int32_t dxh = u * 4 + i - p.nx4;
            if ((int32_t)__fsqrt_rn((float)(dxh * dxh + yz2)) <= srad) (&out.x)[i] = 1.0f;
          }
        }
      }
      vfloat4 ov = {out.x, out.y, out.z, out.w};
      __builtin_nontemporal_store(ov, (vfloat4 *)dcol);
      cm = cc;
      cc = cp;
      col += p.plane;
      dcol += p.plane;
    }
  } else if (u == body4 && (withSphere & 2)) { // head/tail scalar cells (mirrors production)
    for (int32_t z = z0; z < zEnd; ++z) {
      const char *rowC = srcBase + (int64_t)(z + 3) * p.plane + (int64_t)(y + 3) * p.pitch;
      char *rowD = dstBase + (int64_t)(z + 3) * p.plane + (int64_t)(y + 3) * p.pitch;
      for (int32_t lx = 0; lx < 2; ++lx) {
        const int64_t ax = 3 + lx;
        const float v = (*(const float *)(rowC + (ax + 1) * 4) + *(const float *)(rowC + (ax - 1) * 4) +
                         *(const float *)(rowC + p.pitch + ax * 4) + *(const float *)(rowC - p.pitch + ax * 4) +
                         *(const float *)(rowC + p.plane + ax * 4) + *(const float *)(rowC - p.plane + ax * 4)) /
                        6.0f;
        *(float *)(rowD + ax * 4) = v;
      }
    }
  }
}

int main(int argc, char **argv) {
  const int n = argc > 1 ? atoi(argv[1]) : 752;
  const int rounds = argc > 2 ? atoi(argv[2]) : 10;
  P p{};
  p.nx4 = (n - 8) / 4;
  p.ny = n - 6;
  p.nz = n - 6;
  p.pitch = ((int64_t)n * 4 + 255) / 256 * 256;
  p.plane = p.pitch * n;
  char *a, *b;
  CHECK(hipMalloc(&a, p.plane * n));
  CHECK(hipMalloc(&b, p.plane * n));
  CHECK(hipMemset(a, 0x3f, p.plane * n));
  CHECK(hipMemset(b, 0, p.plane * n));
  p.src = a;
  p.dst = b;

  dim3 blk(64, 4, 1);
  dim3 grd((p.nx4 + 63) / 64, (p.ny + 3) / 4, (p.nz + ZCH - 1) / ZCH);
  dim3 grd8((p.nx4 / 2 + 63) / 64, (p.ny + 3) / 4, (p.nz + ZCH - 1) / ZCH);
  hipEvent_t e0, e1;
  CHECK(hipEventCreate(&e0));
  CHECK(hipEventCreate(&e1));
  const double cells = (double)p.nx4 * 4 * p.ny * p.nz;
  // device slots for the production-mirror variants
  char **slots;
  CHECK(hipMalloc(&slots, 2 * sizeof(char *)));
  CHECK(hipMemcpy(slots, &a, sizeof(char *), hipMemcpyHostToDevice));
  CHECK(hipMemcpy(slots + 1, &b, sizeof(char *), hipMemcpyHostToDevice));
  dim3 grdp((p.nx4 + 2 + 63) / 64, (p.ny + 3) / 4, (p.nz + ZCH - 1) / ZCH);

  const int64_t n16 = p.plane * n / 16;
  dim3 grdsw(grd.x, grd.z, grd.y);
  const int NV = 11;
  const char *names[NV] = {"copy", "xz",   "full",    "full8",   "prod-notail",
                           "prod+sph+tail", "prod+tail", "linstream", "full-plainst", "full-dispswap",
                           "full-ldsrows"};
  double best[NV];
  for (int v = 0; v < NV; ++v) best[v] = 1e30;
  for (int r = 0; r < rounds; ++r) {
    for (int v = 0; v < NV; ++v) {
      CHECK(hipEventRecord(e0));
      switch (v) {
      case 0: hipLaunchKernelGGL(probe<0>, grd, blk, 0, 0, p); break;
      case 1: hipLaunchKernelGGL(probe<1>, grd, blk, 0, 0, p); break;
      case 2: hipLaunchKernelGGL(probe<2>, grd, blk, 0, 0, p); break;
      case 7: hipLaunchKernelGGL(probe_lin, dim3(256 * 16), dim3(256), 0, 0, a, b, n16); break;
      case 8: hipLaunchKernelGGL(probe<3>, grd, blk, 0, 0, p); break;
      case 9: hipLaunchKernelGGL(probe_swap, grdsw, blk, 0, 0, p); break;
      case 10: hipLaunchKernelGGL(probe_lds, grd, blk, 0, 0, p); break;
      case 3: hipLaunchKernelGGL(probe8, grd8, blk, 0, 0, p); break;
      case 4:
        hipLaunchKernelGGL(probe_prod, grdp, blk, 0, 0, p, (const char *const *)slots,
                           (char *const *)(slots + 1), 0);
        break;
      case 5:
        hipLaunchKernelGGL(probe_prod, grdp, blk, 0, 0, p, (const char *const *)slots,
                           (char *const *)(slots + 1), 3);
        break;
      case 6:
        hipLaunchKernelGGL(probe_prod, grdp, blk, 0, 0, p, (const char *const *)slots,
                           (char *const *)(slots + 1), 2);
        break;
      }
      CHECK(hipEventRecord(e1));
      CHECK(hipEventSynchronize(e1));
      float ms;
      CHECK(hipEventElapsedTime(&ms, e0, e1));
      best[v] = std::min(best[v], (double)ms);
    }
  }
  for (int v = 0; v < NV; ++v) {
    const double bytes = (v == 7) ? 2.0 * n16 * 16 : cells * 8; // linstream: full alloc R+W
    printf("%-14s %8.3f ms  %7.1f Gcell/s  %6.2f TB/s\n", names[v], best[v],
           cells / best[v] / 1e6, bytes / best[v] / 1e9);
  }
  return 0;
}
