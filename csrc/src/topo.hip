// GPU topology utilities: xGMI link discovery and measured peer bandwidth
// (MI355X-native equivalent of the reference's NVML-based gpu_topo,
// src/gpu_topology.cpp: link distance -> bandwidth for QAP placement).
#include <hip/hip_runtime.h>
#include <hip/hip_ext.h>

#include <vector>

#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/topo.hpp"

namespace stencil_amd {

// distance between two GPUs: 0.1 same device, 1.0 + (hops-1) for
// xGMI-linked peers (all MI355X pairs are one hop), 5.0 no peer access
double gpu_distance(int a, int b) {
  if (a == b) return 0.1;
  int peer = 0;
  if (hipDeviceCanAccessPeer(&peer, a, b) != hipSuccess || !peer) return 5.0;
  uint32_t linkType = 0, hops = 1;
  if (hipExtGetLinkTypeAndHopCount(a, b, &linkType, &hops) == hipSuccess && hops >= 1)
    return 1.0 + (double)(hops - 1);
  return 1.0;
}

// measured unidirectional peer-copy bandwidth (GB/s) via hipMemcpyPeerAsync
double peer_copy_bandwidth(int src, int dst, int64_t bytes, int iters) {
  STENCIL_HIP(hipSetDevice(src));
  hipError_t e = hipDeviceEnablePeerAccess(dst, 0);
  if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) STENCIL_HIP(e);
  (void)hipGetLastError();
  void *sbuf = nullptr, *dbuf = nullptr;
  STENCIL_HIP(hipMalloc(&sbuf, bytes));
  STENCIL_HIP(hipSetDevice(dst));
  e = hipDeviceEnablePeerAccess(src, 0);
  if (e != hipSuccess && e != hipErrorPeerAccessAlreadyEnabled) STENCIL_HIP(e);
  (void)hipGetLastError();
  STENCIL_HIP(hipMalloc(&dbuf, bytes));
  STENCIL_HIP(hipSetDevice(src));
  hipStream_t stream;
  STENCIL_HIP(hipStreamCreateWithFlags(&stream, hipStreamNonBlocking));
  hipEvent_t beg, end;
  STENCIL_HIP(hipEventCreate(&beg));
  STENCIL_HIP(hipEventCreate(&end));
  // warmup
  STENCIL_HIP(hipMemcpyPeerAsync(dbuf, dst, sbuf, src, bytes, stream));
  STENCIL_HIP(hipStreamSynchronize(stream));
  STENCIL_HIP(hipEventRecord(beg, stream));
  for (int i = 0; i < iters; ++i)
    STENCIL_HIP(hipMemcpyPeerAsync(dbuf, dst, sbuf, src, bytes, stream));
  STENCIL_HIP(hipEventRecord(end, stream));
  STENCIL_HIP(hipStreamSynchronize(stream));
  float ms = 0;
  STENCIL_HIP(hipEventElapsedTime(&ms, beg, end));
  (void)hipEventDestroy(beg);
  (void)hipEventDestroy(end);
  (void)hipStreamDestroy(stream);
  (void)hipFree(sbuf);
  STENCIL_HIP(hipSetDevice(dst));
  (void)hipFree(dbuf);
  return (double)bytes * iters / (ms * 1e-3) / 1e9;
}

// device properties for machine-info
GpuInfo gpu_info(int dev) {
  hipDeviceProp_t prop;
  STENCIL_HIP(hipGetDeviceProperties(&prop, dev));
  GpuInfo gi;
  gi.name = prop.name;
  char pci[32];
  snprintf(pci, sizeof(pci), "%04x:%02x:%02x.0", prop.pciDomainID, prop.pciBusID,
           prop.pciDeviceID);
  gi.pci = pci;
  gi.totalMem = (int64_t)prop.totalGlobalMem;
  gi.cuCount = prop.multiProcessorCount;
  return gi;
}

} // namespace stencil_amd
