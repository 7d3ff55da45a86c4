// GPU topology utilities (see csrc/src/topo.hip)
#pragma once

#include <cstdint>
#include <string>

namespace stencil_amd {

double gpu_distance(int a, int b);
double peer_copy_bandwidth(int src, int dst, int64_t bytes, int iters);

struct GpuInfo {
  std::string name;
  std::string pci;
  int64_t totalMem;
  int cuCount;
};
GpuInfo gpu_info(int dev);

} // namespace stencil_amd
