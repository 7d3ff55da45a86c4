// Pure-C++ usage example: the full library without Python — partition,
// placement, planning and transport all come from the C++
// DistributedDomain (distributed.hpp), the same planner the Python path
// uses (reference app shape: bin/jacobi3d.cu).
//
//   build/jacobi3d_native [edge=256] [iters=10] [ngpus=1]
//
// Multi-GPU single process: ngpus > 1 (subdomains split across devices,
// halos move by direct-write xGMI kernels). Multi-process: launch one
// process per GPU with STENCIL_RANK/STENCIL_WORLD/STENCIL_BOOTSTRAP_DIR
// set (see tools/run_native_mp.sh) — cross-rank halos go over RCCL.
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <vector>

#include "stencil_amd/distributed.hpp"
#include "stencil_amd/ops.hpp"

using namespace stencil_amd;

int main(int argc, char **argv) {
  const int64_t n = argc > 1 ? atoll(argv[1]) : 256;
  const int iters = argc > 2 ? atoi(argv[2]) : 10;
  const int ngpus = argc > 3 ? atoi(argv[3]) : 1;

  DistributedDomain dd(n, n, n);
  dd.set_radius(1);
  const int64_t q = dd.add_data<float>("temp");
  if (ngpus > 1) {
    std::vector<int> gpus;
    // STENCIL_FAKE_GPUS=1: put every subdomain on device 0 (the
    // same-GPU fake-multi-GPU test trick, reference test_exchange.cu:52)
    const char *fake = getenv("STENCIL_FAKE_GPUS");
    for (int i = 0; i < ngpus; ++i) gpus.push_back(fake && fake[0] == '1' ? 0 : i);
    dd.set_gpus(gpus);
  }
  dd.realize();

  const Rect3 compute = dd.compute_region();
  const auto interiors = dd.get_interior();
  const auto exteriors = dd.get_exterior();
  for (int li = 0; li < dd.num_local(); ++li) {
    fill_f32(dd.engine(), li, q, dd.local_rect(li), 0.5f, false);
    fill_f32(dd.engine(), li, q, dd.local_rect(li), 0.5f, true);
  }
  dd.engine().sync_compute();

  const auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it) {
    // interior compute overlaps the halo exchange
    for (int li = 0; li < dd.num_local(); ++li)
      jacobi_step(dd.engine(), li, q, interiors[li], compute);
    dd.exchange();
    for (int li = 0; li < dd.num_local(); ++li)
      for (const Rect3 &box : exteriors[li])
        jacobi_step(dd.engine(), li, q, box, compute, /*streamId=*/1);
    dd.engine().sync_compute();
    dd.swap();
  }
  const std::chrono::duration<double> dt = std::chrono::steady_clock::now() - t0;
  if (getenv("STENCIL_PARAVIEW")) dd.write_paraview("jacobi_pv_");

  if (dd.rank() == 0)
    printf("jacobi3d_native: %lld^3 world=%d gpus=%d, %d iters, %.3f ms/iter, %.1f Gcell/s, "
           "%lld translate B + %lld wire B per exchange\n",
           (long long)n, dd.world(), ngpus, iters, dt.count() / iters * 1e3,
           (double)n * n * n * iters / dt.count() / 1e9, (long long)dd.bytes_translate(),
           (long long)dd.bytes_wire());
  return 0;
}
