// Jacobi-3D 7-point stencil kernels for gfx950 (the flagship app workload;
// reference: bin/jacobi3d.cu:18-85).
//
// The kernel is HBM-bandwidth-bound: per cell it reads the 6 face neighbors
// + writes one value; with y/z-neighbor rows served from L2/L3 the traffic
// floor is 4 B read + 4 B write per cell (PMC-verified: 1.64 GB fetched /
// 1.75 GB written per 750^3 step, profiles/jacobi_750_*_size.csv).
// Design choices for CDNA4, each A/B-measured via examples/jacobi_probe:
//   - float4 x-strips marching z with register-rolled center planes,
//     nontemporal stores (plain stores 9% slower),
//   - jacobi_kernel_v4_lds (the vecAll/graph path): y-neighbor rows
//     staged through 12 KB of ping-pong LDS, full 8 waves/SIMD — runs at
//     the mapping's copy roofline (540 Gcell/s @750^3),
//   - pure-vector full-rect launches (tail lanes cost 10%) with
//     address-based 16 B alignment,
//   - linearized scalar kernel for thin exterior slabs (any shape stays
//     coalesced), grid-stride with a capped grid.
// Hot/cold sphere sources match the reference's behavior (truncated-int
// sqrtf distance, HOT=1/COLD=0 fixed cells) so results are comparable.
#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstdio>
#include <cstdlib>
#include <memory>
#include <vector>

#include "stencil_amd/device_util.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"

namespace stencil_amd {

namespace {

struct JacobiParams {
  // raw base pointers passed per launch (kernarg pointers get global
  // address-space inference + SGPR preload; slot-indirected bases cost
  // ~18% on this kernel -- measured with the jacobi_probe prod variant)
  const char *src;
  char *dst;
  int64_t pitch, plane;       // byte strides (same for curr/next)
  // global coordinate of allocation element (0,0,0)
  int64_t allocX, allocY, allocZ;
  // region to compute, global coords
  int64_t loX, loY, loZ;
  int32_t extX, extY, extZ;
  int32_t vecAll; // 1 = pure-vector launch (see launch_jacobi_on)
  // whole compute region (for the hot/cold spheres)
  int64_t cLoX, cLoY, cLoZ, cHiX, cHiY, cHiZ;
};

__device__ __forceinline__ bool sphere_override(int64_t x, int64_t y, int64_t z,
                                                const JacobiParams &p, float &out) {
  const int64_t cw = p.cHiX - p.cLoX;
  const int64_t hotX = p.cLoX + cw / 3, coldX = p.cLoX + cw * 2 / 3;
  const int64_t cY = (p.cLoY + p.cHiY) / 2, cZ = (p.cLoZ + p.cHiZ) / 2;
  const int64_t r = cw / 10;
  {
    const int64_t dx = x - hotX, dy = y - cY, dz = z - cZ;
    const int64_t d = (int64_t)__fsqrt_rn((float)(dx * dx + dy * dy + dz * dz));
    if (d <= r) {
      out = 1.0f;
      return true;
    }
  }
  {
    const int64_t dx = x - coldX, dy = y - cY, dz = z - cZ;
    const int64_t d = (int64_t)__fsqrt_rn((float)(dx * dx + dy * dy + dz * dz));
    if (d <= r) {
      out = 0.0f;
      return true;
    }
  }
  return false;
}

// Vectorized z-marching kernel for wide regions: each thread owns a
// 4-wide x strip (16 B aligned) and marches ZCHUNK cells in z, keeping the
// z-1 / z / z+1 center vectors in registers so the +-z neighbor planes are
// never re-loaded. Per 4 cells and z step: 3 aligned float4 loads (new
// center plane + y neighbors) + 2 scalar edge loads + 1 float4 store
// (~14 B/cell issued vs 28 for the scalar kernel). The hot/cold sphere
// test short-circuits on a 1D bounding check so the common case costs two
// int compares.
#define JAC_ZCHUNK 16

__global__ void __launch_bounds__(256) jacobi_kernel_v4(JacobiParams p) {
  // (an XCD-aware block remap was measured 9% SLOWER here: the row
  // working set is L3-resident, so the remap only disturbed dispatch)
  const int32_t u = blockIdx.x * blockDim.x + threadIdx.x; // x unit
  const int32_t ly = blockIdx.y * blockDim.y + threadIdx.y;
  const int32_t lz0 = blockIdx.z * JAC_ZCHUNK;
  if (ly >= p.extY) return;
  const char *srcBase = p.src;
  char *dstBase = p.dst;
  const int64_t gy = p.loY + ly;
  const int64_t ay = gy - p.allocY;
  const int32_t zEnd = min((int32_t)(lz0 + JAC_ZCHUNK), p.extZ);

  // sphere geometry (int32: grids are far below 2^31 per axis)
  const int32_t cw = (int32_t)(p.cHiX - p.cLoX);
  const int32_t hotX = (int32_t)p.cLoX + cw / 3, coldX = (int32_t)p.cLoX + cw * 2 / 3;
  const int32_t cY = (int32_t)(p.cLoY + p.cHiY) / 2, cZ = (int32_t)(p.cLoZ + p.cHiZ) / 2;
  const int32_t srad = cw / 10;

  auto sphere4 = [&](int32_t gx, int32_t gyy, int32_t gzz, float4 &out) {
    const int32_t dy = gyy - cY, dz = gzz - cZ;
    const int32_t yz2 = dy * dy + dz * dz;
    // truncated-int sqrt: only yz2 >= (r+1)^2 guarantees exclusion
    if (yz2 >= (srad + 1) * (srad + 1)) return;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int32_t dxh = gx + i - hotX, dxc = gx + i - coldX;
      if ((int32_t)__fsqrt_rn((float)(dxh * dxh + yz2)) <= srad)
        (&out.x)[i] = 1.0f;
      else if ((int32_t)__fsqrt_rn((float)(dxc * dxc + yz2)) <= srad)
        (&out.x)[i] = 0.0f;
    }
  };

  auto scalar_cell = [&](int32_t lx, int32_t lz) {
    const int64_t gx = p.loX + lx;
    const int64_t gz = p.loZ + lz;
    const int64_t ax = gx - p.allocX, az = gz - p.allocZ;
    const char *rowC = srcBase + az * p.plane + ay * p.pitch;
    float out;
    if (!sphere_override(gx, gy, gz, p, out)) {
      const float px = *(const float *)(rowC + (ax + 1) * 4);
      const float mx = *(const float *)(rowC + (ax - 1) * 4);
      const float py = *(const float *)(rowC + p.pitch + ax * 4);
      const float my = *(const float *)(rowC - p.pitch + ax * 4);
      const float pz = *(const float *)(rowC + p.plane + ax * 4);
      const float mz = *(const float *)(rowC - p.plane + ax * 4);
      out = (px + mx + py + my + pz + mz) / 6.0f;
    }
    *(float *)(dstBase + az * p.plane + ay * p.pitch + ax * 4) = out;
  };

  const int64_t a0 = p.loX - p.allocX;
  // vecAll: host pre-adjusted loX/extX (16B-aligned start, extX%4==0,
  // writes beyond the true region land only in halo/row-slack bytes that
  // the next exchange refreshes before any read) -- no scalar lanes.
  // Otherwise head = elements until the next 16 B-aligned ADDRESS (the
  // allocation pad shifts the base, so index-aligned units would issue
  // misaligned dwordx4 -- measured ~10% on gfx950); must match the
  // launcher's units formula.
  const int32_t headAl =
      (int32_t)((16 - (((uintptr_t)srcBase + (uintptr_t)(a0 * 4)) & 15)) & 15) >> 2;
  const int32_t head = p.vecAll ? 0 : (headAl > p.extX ? p.extX : headAl);
  const int32_t body4 = (p.extX - head) / 4;
  const int32_t tail = p.extX - head - body4 * 4;

  if (u < body4) {
    const int64_t ax = a0 + head + (int64_t)u * 4; // 4-aligned
    const int32_t gx = (int32_t)(ax + p.allocX);
    const int64_t az0 = p.loZ + lz0 - p.allocZ;
    const char *col = srcBase + az0 * p.plane + ay * p.pitch + ax * 4;
    char *dcol = dstBase + az0 * p.plane + ay * p.pitch + ax * 4;
    // rolling center vectors
    float4 cm = *(const float4 *)(col - p.plane);
    float4 cc = *(const float4 *)(col);
    for (int32_t lz = lz0; lz < zEnd; ++lz) {
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
      sphere4(gx, (int32_t)gy, (int32_t)(p.loZ + lz), out);
      // next is write-only this iteration: bypass cache pollution
      typedef float vfloat4 __attribute__((ext_vector_type(4)));
      vfloat4 ov = {out.x, out.y, out.z, out.w};
      __builtin_nontemporal_store(ov, (vfloat4 *)dcol);
      cm = cc;
      cc = cp;
      col += p.plane;
      dcol += p.plane;
    }
  } else if (u == body4) {
    // (spreading these cells over one-lane-per-z-plane measured 10%
    // SLOWER overall -- the extra block column cost more than the
    // straggler lane; keep the compact form)
    for (int32_t lz = lz0; lz < zEnd; ++lz)
      for (int32_t lx = 0; lx < head; ++lx) scalar_cell(lx, lz);
  } else if (u == body4 + 1) {
    for (int32_t lz = lz0; lz < zEnd; ++lz)
      for (int32_t lx = p.extX - tail; lx < p.extX; ++lx) scalar_cell(lx, lz);
  }
}

// vecAll-only variant: y-neighbor rows staged through LDS. A 64x4 block
// marching z keeps the current plane's center vectors of its 4 rows + 2
// y-halo rows in ping-pong LDS buffers (one barrier per z-step), so the
// py/my global vector loads become LDS reads. Probe: 0.749 ms vs 0.839
// for the pure-global stencil at 752^3 -- at the mapping's copy roofline
// (examples/jacobi_probe.cpp full-ldsrows). Requires blockDim (64,4,1)
// and a vecAll launch (no scalar lanes; every lane's row exists because
// the region is the interior rect of a radius>=1 domain).
template <int ZC>
__global__ void __launch_bounds__(256) jacobi_kernel_v4_lds(JacobiParams p) {
  __shared__ float4 tile[2][6][64];
  const int32_t tx = threadIdx.x;
  const int32_t ry = threadIdx.y;
  // no early returns (__syncthreads needs every lane): out-of-range lanes
  // clamp coordinates and skip only the store
  const int32_t u0 = blockIdx.x * 64 + tx;
  const int32_t ly0 = blockIdx.y * 4 + ry;
  const int32_t body4 = p.extX / 4; // vecAll: extX % 4 == 0
  const bool valid = u0 < body4 && ly0 < p.extY;
  const int32_t u = min(u0, body4 - 1);
  // clamp to extY (the +y halo row, which exists for radius >= 1): a
  // VALID lane ry-1 reads lane ry's staged row as its +y neighbor, so
  // lanes at ly0 == extY must stage the TRUE halo row, not a duplicate
  const int32_t ly = min(ly0, (int32_t)p.extY);
  // +y halo row staged by ry==3: one past its own row, clamped in-bounds
  const int64_t pyHalo = (ly < p.extY) ? p.pitch : 0;
  const int32_t lz0 = blockIdx.z * ZC;
  const int32_t zEnd = min((int32_t)(lz0 + ZC), p.extZ);

  const int64_t gy = p.loY + ly;
  const int64_t ay = gy - p.allocY;
  const int64_t a0 = p.loX - p.allocX;
  const int64_t ax = a0 + (int64_t)u * 4;
  const int32_t gx = (int32_t)(ax + p.allocX);
  const int64_t az0 = p.loZ + lz0 - p.allocZ;
  const char *col = p.src + az0 * p.plane + ay * p.pitch + ax * 4;
  char *dcol = p.dst + az0 * p.plane + ay * p.pitch + ax * 4;

  const int32_t cw = (int32_t)(p.cHiX - p.cLoX);
  const int32_t hotX = (int32_t)p.cLoX + cw / 3, coldX = (int32_t)p.cLoX + cw * 2 / 3;
  const int32_t cY = (int32_t)(p.cLoY + p.cHiY) / 2, cZ = (int32_t)(p.cLoZ + p.cHiZ) / 2;
  const int32_t srad = cw / 10;
  auto sphere4 = [&](int32_t gzz, float4 &out) {
    const int32_t dy = (int32_t)gy - cY, dz = gzz - cZ;
    const int32_t yz2 = dy * dy + dz * dz;
    if (yz2 >= (srad + 1) * (srad + 1)) return;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      const int32_t dxh = gx + i - hotX, dxc = gx + i - coldX;
      if ((int32_t)__fsqrt_rn((float)(dxh * dxh + yz2)) <= srad)
        (&out.x)[i] = 1.0f;
      else if ((int32_t)__fsqrt_rn((float)(dxc * dxc + yz2)) <= srad)
        (&out.x)[i] = 0.0f;
    }
  };

  float4 cm = *(const float4 *)(col - p.plane);
  float4 cc = *(const float4 *)(col);
  tile[0][ry + 1][tx] = cc;
  if (ry == 0) tile[0][0][tx] = *(const float4 *)(col - p.pitch);
  if (ry == 3) tile[0][5][tx] = *(const float4 *)(col + pyHalo);
  __syncthreads();
  int buf = 0;
  for (int32_t lz = lz0; lz < zEnd; ++lz) {
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
    sphere4((int32_t)(p.loZ + lz), out);
    if (valid) {
      typedef float vfloat4 __attribute__((ext_vector_type(4)));
      vfloat4 ov = {out.x, out.y, out.z, out.w};
      __builtin_nontemporal_store(ov, (vfloat4 *)dcol);
    }
    tile[buf ^ 1][ry + 1][tx] = cp;
    if (ry == 0) tile[buf ^ 1][0][tx] = *(const float4 *)(col + p.plane - p.pitch);
    if (ry == 3) tile[buf ^ 1][5][tx] = *(const float4 *)(col + p.plane + pyHalo);
    __syncthreads();
    buf ^= 1;
    cm = cc;
    cc = cp;
    col += p.plane;
    dcol += p.plane;
  }
}

__global__ void jacobi_kernel(JacobiParams p) {
  const char *srcBase = p.src;
  char *dstBase = p.dst;
  const int64_t total = (int64_t)p.extX * p.extY * p.extZ;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const int32_t lx = (int32_t)(i % p.extX);
    const int64_t t = i / p.extX;
    const int32_t ly = (int32_t)(t % p.extY);
    const int32_t lz = (int32_t)(t / p.extY);
    // global coords
    const int64_t gx = p.loX + lx, gy = p.loY + ly, gz = p.loZ + lz;
    // allocation-local element coords
    const int64_t ax = gx - p.allocX, ay = gy - p.allocY, az = gz - p.allocZ;
    const char *rowC = srcBase + az * p.plane + ay * p.pitch;
    float out;
    if (!sphere_override(gx, gy, gz, p, out)) {
      const float px = *(const float *)(rowC + (ax + 1) * 4);
      const float mx = *(const float *)(rowC + (ax - 1) * 4);
      const float py = *(const float *)(rowC + p.pitch + ax * 4);
      const float my = *(const float *)(rowC - p.pitch + ax * 4);
      const float pz = *(const float *)(rowC + p.plane + ax * 4);
      const float mz = *(const float *)(rowC - p.plane + ax * 4);
      out = (px + mx + py + my + pz + mz) / 6.0f;
    }
    *(float *)(dstBase + az * p.plane + ay * p.pitch + ax * 4) = out;
  }
}

__global__ void fill_kernel(char *base, int64_t pitch, int64_t plane,
                            int32_t extX, int32_t extY, int32_t extZ, float value) {
  const int64_t total = (int64_t)extX * extY * extZ;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; i < total; i += stride) {
    const int32_t lx = (int32_t)(i % extX);
    const int64_t t = i / extX;
    const int32_t ly = (int32_t)(t % extY);
    const int64_t lz = t / extY;
    *(float *)(base + lz * plane + (int64_t)ly * pitch + (int64_t)lx * 4) = value;
  }
}

uint32_t grid_for(int64_t total, int block) {
  int64_t g = (total + block - 1) / block;
  // cap: 256 CUs x 16 blocks keeps the chip full while bounding launch size
  g = std::min<int64_t>(g, 256 * 16);
  return (uint32_t)std::max<int64_t>(g, 1);
}

} // namespace

namespace {

// shared launch logic: builds params from the domain's CURRENT buffer
// parity and enqueues the right kernel variant onto `stream`. Also used
// by the whole-step hipGraph capture (pointers get baked per parity).
// fullRectVec: 0 = off; 1 = free extension (single-process or RCCL-wire
// ranks: extended writes land in refresh-before-read halo/slack bytes or
// exterior cells a later same-stream kernel rewrites); 2 = strict (IPC
// ranks: a fast peer writes our NEXT-buffer halos during our compute, so
// no extension is allowed -- the fast path engages only when the region
// is already aligned, which the allocation pad arranges for the overlap
// interior).
void launch_jacobi_on(LocalDomain &d, int64_t qi, const Rect3 &region,
                      const Rect3 &computeRegion, hipStream_t stream, int fullRectVec = 0) {
  if (d.elem_size(qi) != 4) throw std::runtime_error("jacobi_step: quantity must be fp32");
  Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  JacobiParams p{};
  p.src = d.curr(qi).ptr;
  p.dst = d.next(qi).ptr;
  p.pitch = d.curr(qi).pitch;
  p.plane = d.curr(qi).plane();
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
  p.cLoX = computeRegion.lo.x;
  p.cLoY = computeRegion.lo.y;
  p.cLoZ = computeRegion.lo.z;
  p.cHiX = computeRegion.hi.x;
  p.cHiY = computeRegion.hi.y;
  p.cHiZ = computeRegion.hi.z;
  // fullRectVec: the region is the whole interior rect (radius >= 1 on
  // both x sides). Extend it left to the previous 16 B-aligned ADDRESS
  // and right to a multiple of 4, eliminating the scalar head/tail lanes
  // (the probe measured them at ~10% of the kernel). The extended cells
  // are x-halo cells or pitch-slack bytes of the adjacent row: every one
  // is rewritten by the next exchange (or is padding) before any read,
  // and never a translate SOURCE (sources are interior cells).
  if (fullRectVec && ext.x >= 8) {
    const int64_t a0 = region.lo.x - full.lo.x;
    const uintptr_t addr0 = (uintptr_t)p.src + a0 * 4;
    const int64_t under = (int64_t)((addr0 & 15) >> 2);
    int64_t e2 = ext.x + under;
    const int64_t over = (4 - (e2 & 3)) & 3;
    e2 += over;
    if (fullRectVec != 2 || (under == 0 && over == 0)) {
      p.loX = region.lo.x - under;
      p.extX = (int32_t)e2;
      p.vecAll = 1;
      ext.x = e2;
    }
  }
  static int useLds = -1;
  if (useLds < 0) {
    const char *e = getenv("STENCIL_JAC_LDS");
    useLds = (e && e[0] == '0') ? 0 : 1;
  }
  if (p.vecAll && useLds && ext.x >= 8) {
    // LDS-staged y-rows variant (fixed 64x4 block); see jacobi_kernel_v4_lds
    static int zc = 0;
    if (!zc) {
      const char *e = getenv("STENCIL_JAC_ZC");
      zc = (e && atoi(e) == 16) ? 16 : 32; // 32 measured +1.8% (gpu35)
    }
    dim3 block(64, 4, 1);
    dim3 grid((uint32_t)((p.extX / 4 + 63) / 64), (uint32_t)((ext.y + 3) / 4),
              (uint32_t)((ext.z + zc - 1) / zc));
    if (zc == 32)
      hipLaunchKernelGGL(jacobi_kernel_v4_lds<32>, grid, block, 0, stream, p);
    else
      hipLaunchKernelGGL(jacobi_kernel_v4_lds<16>, grid, block, 0, stream, p);
    STENCIL_HIP(hipGetLastError());
    return;
  }
  if (ext.x >= 8 && ext.y <= 0x7fffffff) {
    // vectorized row-mapped kernel; block shape tunable via env
    const int64_t a0 = p.loX - full.lo.x;
    const int64_t headAl =
        (int64_t)((16 - (((uintptr_t)p.src + (uintptr_t)(a0 * 4)) & 15)) & 15) >> 2;
    const int64_t head = p.vecAll ? 0 : std::min<int64_t>(headAl, ext.x);
    const int64_t units = (ext.x - head) / 4 + (p.vecAll ? 0 : 2);
    static int bx = 0, by = 0;
    if (!bx) {
      bx = 64;
      by = 4;
      if (const char *e = getenv("STENCIL_JAC_BLOCK")) {
        if (sscanf(e, "%dx%d", &bx, &by) != 2 || bx * by != 256) {
          bx = 64;
          by = 4;
        }
      }
    }
    dim3 block((uint32_t)bx, (uint32_t)by, 1);
    dim3 grid((uint32_t)((units + bx - 1) / bx), (uint32_t)((ext.y + by - 1) / by),
              (uint32_t)((ext.z + 15) / 16)); // 16 == JAC_ZCHUNK
    hipLaunchKernelGGL(jacobi_kernel_v4, grid, block, 0, stream, p);
  } else {
    hipLaunchKernelGGL(jacobi_kernel, dim3(grid_for(ext.flatten(), 256)), dim3(256), 0,
                       stream, p);
  }
  STENCIL_HIP(hipGetLastError());
}

} // namespace

void jacobi_step(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                 const Rect3 &computeRegion, int streamId, int extendVec) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  launch_jacobi_on(d, qi, region, computeRegion, eng.compute_stream(dom, streamId),
                   extendVec);
}

namespace {

// One graph per buffer parity: the jacobi kernargs are baked per parity
// (slot-indirecting the jacobi kernel itself was measured ~18% slower),
// while the translate jobs stay slot-indirect and an in-graph
// LocalDomain::enqueue_table_swap node flips the device tables.
struct StepGraph {
  hipStream_t stream = nullptr;
  hipGraphExec_t exec[2] = {nullptr, nullptr};
  int parity = 0;
  ExchangeEngine *eng = nullptr;
  int dom = 0;
};
std::vector<std::unique_ptr<StepGraph>> g_stepGraphs;

} // namespace

int64_t jacobi_graph_create(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                            const Rect3 &computeRegion) {
  // whole-step replay graph: [translate copy_batch -> full-region jacobi
  // -> device-side table swap] captured once per buffer parity. Replay
  // costs one hipGraphLaunch (~5 us) instead of the ~0.25 ms of host
  // orchestration measured per step at 750^3 (exchange + launches +
  // stream syncs + swap upload). Single-process single-domain only (the
  // periodic self-wrap bench shape): no wire/IPC machinery may exist.
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  auto sg = std::make_unique<StepGraph>();
  sg->eng = &eng;
  sg->dom = dom;
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream, hipStreamNonBlocking));
  for (int par = 0; par < 2; ++par) {
    STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
    eng.launch_translates_plain_on((uintptr_t)sg->stream, 0);
    // the graph's region is the whole interior rect: pure-vector launch
    // (scalar tail lanes measured ~10% of the kernel in the probe)
    launch_jacobi_on(d, qi, region, computeRegion, sg->stream, /*fullRectVec=*/1);
    d.enqueue_table_swap(sg->stream);
    hipGraph_t g = nullptr;
    STENCIL_HIP(hipStreamEndCapture(sg->stream, &g));
    STENCIL_HIP(hipGraphInstantiate(&sg->exec[par], g, nullptr, nullptr, 0));
    STENCIL_HIP(hipGraphDestroy(g));
    d.swap(); // bake the other parity's kernarg pointers next round
  }
  // two swaps: host+device state is back where it started
  g_stepGraphs.push_back(std::move(sg));
  return (int64_t)g_stepGraphs.size() - 1;
}

// Overlap variant of the whole-step graph (STENCIL_AMD_GRAPH_OVERLAP=1):
// fork the translate copy_batch onto a second stream concurrent with the
// interior kernel, join, then exterior shells + swap. The round-1 eager
// A/B called overlap "a wash", but that measured per-step host
// orchestration against the SERIAL graph; in-graph the fork costs
// nothing, so the ~60 us copy_batch can hide under the ~700 us interior.
// Interior runs fullRectVec mode 1 (free extension); its x-overshoot
// lands in cells the +-x exterior slabs rewrite LATER ON THE SAME
// STREAM, which is exactly the eager-mode ordering argument.
int64_t jacobi_graph_create_overlap(ExchangeEngine &eng, int dom, int64_t qi,
                                    const Rect3 &interior, const Rect3 &computeRegion,
                                    const std::vector<Rect3> &exteriors) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  auto sg = std::make_unique<StepGraph>();
  sg->eng = &eng;
  sg->dom = dom;
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream, hipStreamNonBlocking));
  hipStream_t stream2;
  STENCIL_HIP(hipStreamCreateWithFlags(&stream2, hipStreamNonBlocking));
  hipEvent_t evF, evJ;
  STENCIL_HIP(hipEventCreateWithFlags(&evF, hipEventDisableTiming));
  STENCIL_HIP(hipEventCreateWithFlags(&evJ, hipEventDisableTiming));
  for (int par = 0; par < 2; ++par) {
    STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
    STENCIL_HIP(hipEventRecord(evF, sg->stream));
    STENCIL_HIP(hipStreamWaitEvent(stream2, evF, 0));
    eng.launch_translates_plain_on((uintptr_t)stream2, 0); // halos || interior
    launch_jacobi_on(d, qi, interior, computeRegion, sg->stream, /*fullRectVec=*/1);
    STENCIL_HIP(hipEventRecord(evJ, stream2));
    STENCIL_HIP(hipStreamWaitEvent(sg->stream, evJ, 0));
    for (const Rect3 &box : exteriors)
      launch_jacobi_on(d, qi, box, computeRegion, sg->stream, /*fullRectVec=*/0);
    d.enqueue_table_swap(sg->stream);
    hipGraph_t g = nullptr;
    STENCIL_HIP(hipStreamEndCapture(sg->stream, &g));
    STENCIL_HIP(hipGraphInstantiate(&sg->exec[par], g, nullptr, nullptr, 0));
    STENCIL_HIP(hipGraphDestroy(g));
    d.swap();
  }
  STENCIL_HIP(hipStreamDestroy(stream2));
  STENCIL_HIP(hipEventDestroy(evF));
  STENCIL_HIP(hipEventDestroy(evJ));
  g_stepGraphs.push_back(std::move(sg));
  return (int64_t)g_stepGraphs.size() - 1;
}

void jacobi_graph_launch(int64_t handle, int64_t nSteps) {
  StepGraph &sg = *g_stepGraphs.at(handle);
  LocalDomain &d = sg.eng->domain(sg.dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  for (int64_t i = 0; i < nSteps; ++i) {
    STENCIL_HIP(hipGraphLaunch(sg.exec[sg.parity], sg.stream));
    sg.parity ^= 1;
    d.swap_host_only(); // in-graph kernel flips the device tables
  }
}

void jacobi_graph_sync(int64_t handle) {
  StepGraph &sg = *g_stepGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipStreamSynchronize(sg.stream));
}

// Multi-rank whole-step graphs (one rank, one domain, cross-rank halos
// via IPC direct writes -- the 8-GPU single-node bench shape). The step
// splits at the cross-rank barrier:
//   A[par] = [interior jacobi -> translates (self + IPC views) ->
//             staged thin packs (parity)]
//   <barrier: all ranks' A complete -- RCCL all-reduce posted on the
//    SAME stream (fully stream-ordered), or a gloo host barrier with a
//    stream sync around it>
//   B[par] = [staged unpacks (parity) -> exterior slabs -> device table
//             swap -> device view flips]
// With the device barrier, steps queue back-to-back with ONE host sync
// per run(n): the eager path's ~0.21 ms/step of host orchestration
// (measured 0.989 vs 0.780 ms, profiles/r2/r2_gpu7_eager.log) collapses
// to two graph launches + one collective post per step.
namespace {
struct MrStepGraph {
  hipStream_t stream = nullptr;
  hipGraphExec_t execA[2] = {nullptr, nullptr};
  hipGraphExec_t execB[2] = {nullptr, nullptr};
  int parity = 0;
  ExchangeEngine *eng = nullptr;
  int dom = 0;
};
std::vector<std::unique_ptr<MrStepGraph>> g_mrGraphs;
} // namespace

int64_t jacobi_mr_graph_create(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &interior,
                               const Rect3 &computeRegion, const std::vector<Rect3> &exteriors,
                               int extendVec) {
  LocalDomain &d = eng.domain(dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  auto sg = std::make_unique<MrStepGraph>();
  sg->eng = &eng;
  sg->dom = dom;
  STENCIL_HIP(hipStreamCreateWithFlags(&sg->stream, hipStreamNonBlocking));
  for (int par = 0; par < 2; ++par) {
    // A: interior + outgoing halos
    STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
    launch_jacobi_on(d, qi, interior, computeRegion, sg->stream, extendVec);
    eng.launch_translates_plain_on((uintptr_t)sg->stream, 0);
    eng.launch_packs_plain_on((uintptr_t)sg->stream, 1 + par); // staged parity
    hipGraph_t ga = nullptr;
    STENCIL_HIP(hipStreamEndCapture(sg->stream, &ga));
    STENCIL_HIP(hipGraphInstantiate(&sg->execA[par], ga, nullptr, nullptr, 0));
    STENCIL_HIP(hipGraphDestroy(ga));
    // B: incoming halos + exterior + swap
    STENCIL_HIP(hipStreamBeginCapture(sg->stream, hipStreamCaptureModeThreadLocal));
    eng.launch_unpacks_plain_on((uintptr_t)sg->stream, 1 + par);
    for (const Rect3 &box : exteriors)
      launch_jacobi_on(d, qi, box, computeRegion, sg->stream, /*fullRectVec=*/0);
    d.enqueue_table_swap(sg->stream);
    eng.enqueue_view_flips((uintptr_t)sg->stream);
    hipGraph_t gb = nullptr;
    STENCIL_HIP(hipStreamEndCapture(sg->stream, &gb));
    STENCIL_HIP(hipGraphInstantiate(&sg->execB[par], gb, nullptr, nullptr, 0));
    STENCIL_HIP(hipGraphDestroy(gb));
    d.swap(); // bake the other parity's kernarg pointers next round
  }
  g_mrGraphs.push_back(std::move(sg));
  return (int64_t)g_mrGraphs.size() - 1;
}

uintptr_t jacobi_mr_graph_stream(int64_t handle) {
  return (uintptr_t)g_mrGraphs.at(handle)->stream;
}

void jacobi_mr_graph_pre(int64_t handle) {
  MrStepGraph &sg = *g_mrGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipGraphLaunch(sg.execA[sg.parity], sg.stream));
}

void jacobi_mr_graph_post(int64_t handle) {
  MrStepGraph &sg = *g_mrGraphs.at(handle);
  LocalDomain &d = sg.eng->domain(sg.dom);
  STENCIL_HIP(hipSetDevice(d.gpu()));
  STENCIL_HIP(hipGraphLaunch(sg.execB[sg.parity], sg.stream));
  sg.parity ^= 1;
  d.swap_host_only(); // in-graph kernels flip the device state
  sg.eng->flip_views_host_only();
}

void jacobi_mr_graph_sync(int64_t handle) {
  MrStepGraph &sg = *g_mrGraphs.at(handle);
  STENCIL_HIP(hipSetDevice(sg.eng->domain(sg.dom).gpu()));
  STENCIL_HIP(hipStreamSynchronize(sg.stream));
}

void fill_f32(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, float value,
              bool nextBuf) {
  LocalDomain &d = eng.domain(dom);
  if (d.elem_size(qi) != 4) throw std::runtime_error("fill_f32: quantity must be fp32");
  const Vec3 ext = region.extent();
  if (ext.flatten() <= 0) return;
  const Rect3 full = d.full_region();
  const Vec3 pos = region.lo - full.lo; // allocation coords
  const Pitched &pp = d.curr(qi);
  const int64_t off = pos.z * pp.plane() + pos.y * pp.pitch + pos.x * 4;
  char *base = (nextBuf ? d.next(qi).ptr : d.curr(qi).ptr) + off;
  STENCIL_HIP(hipSetDevice(d.gpu()));
  hipLaunchKernelGGL(fill_kernel, dim3(grid_for(ext.flatten(), 256)), dim3(256), 0,
                     eng.compute_stream(dom), base, pp.pitch, pp.plane(), (int32_t)ext.x,
                     (int32_t)ext.y, (int32_t)ext.z, value);
  STENCIL_HIP(hipGetLastError());
}

} // namespace stencil_amd
