// Pure-C++ usage example: the framework without Python. One process, one
// (or more) GPUs; periodic self-exchange; jacobi iterations with
// interior/exterior overlap. Build: tools/build_native.py --with-examples
// -> build/jacobi3d_native.
#include <chrono>
#include <cstdio>
#include <memory>
#include <vector>

#include "stencil_amd/core.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"
#include "stencil_amd/partition.hpp"

using namespace stencil_amd;

int main(int argc, char **argv) {
  const int64_t n = argc > 1 ? atoll(argv[1]) : 256;
  const int iters = argc > 2 ? atoi(argv[2]) : 10;
  const Radius radius = Radius::constant(1);

  auto dom = std::make_shared<LocalDomain>(Vec3(n, n, n), Vec3(0, 0, 0), 0);
  dom->set_radius(radius);
  const int64_t q = dom->add_data(sizeof(float), "temp");
  dom->realize();

  ExchangeEngine eng({dom});
  // periodic self-exchange: all 26 directions wrap onto the same domain
  for (int dz = -1; dz <= 1; ++dz)
    for (int dy = -1; dy <= 1; ++dy)
      for (int dx = -1; dx <= 1; ++dx) {
        const Vec3 d(dx, dy, dz);
        if (d == Vec3(0, 0, 0) || radius.dir(-d) == 0) continue;
        eng.add_translate(0, 0, dom->halo_pos(d, false), dom->halo_pos(-d, true),
                          LocalDomain::halo_extent(-d, dom->size(), radius));
      }
  eng.finalize();

  const Rect3 compute = dom->compute_region();
  const Rect3 interior(compute.lo + Vec3(1, 1, 1), compute.hi - Vec3(1, 1, 1));
  fill_f32(eng, 0, q, compute, 0.5f, false);
  fill_f32(eng, 0, q, compute, 0.5f, true);
  eng.sync_compute();

  const auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it) {
    jacobi_step(eng, 0, q, interior, compute); // overlaps with...
    eng.launch_translates();                   // ...the halo exchange
    eng.sync_translates();
    // exterior shells on the second compute stream: they depend only on
    // the (synced) exchange, so they overlap the interior kernel
    // (slide faces in; reference src/stencil.cu:927-977)
    Rect3 c = compute;
    for (int axis = 0; axis < 3; ++axis) {
      Rect3 s = c;
      s.lo[axis] = interior.hi[axis];
      jacobi_step(eng, 0, q, s, compute, /*streamId=*/1);
      c.hi[axis] = interior.hi[axis];
      Rect3 t = c;
      t.hi[axis] = interior.lo[axis];
      jacobi_step(eng, 0, q, t, compute, /*streamId=*/1);
      c.lo[axis] = interior.lo[axis];
    }
    eng.sync_compute();
    dom->swap();
  }
  const std::chrono::duration<double> dt = std::chrono::steady_clock::now() - t0;
  printf("jacobi3d_native: %lld^3, %d iters, %.3f ms/iter, %.1f Gcell/s\n", (long long)n, iters,
         dt.count() / iters * 1e3, (double)n * n * n * iters / dt.count() / 1e9);
  return 0;
}
