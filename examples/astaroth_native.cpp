// Pure-C++ usage example: the Astaroth-class MHD solver without Python
// (reference astaroth/astaroth.cu is C++-only; this is the equivalent
// minimal driver on the stencil_amd C++ API). One process, one GPU,
// periodic self-exchange, separable-derivative RK3 with quantity-group
// exchanges. Build: tools/build_native.py -> build/astaroth_native.
#include <chrono>
#include <cmath>
#include <cstdio>
#include <memory>
#include <vector>

#include "stencil_amd/core.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"

using namespace stencil_amd;

int main(int argc, char **argv) {
  const int64_t n = argc > 1 ? atoll(argv[1]) : 128;
  const int iters = argc > 2 ? atoi(argv[2]) : 5;
  const Radius radius = Radius::constant(3);
  const double dt = 1e-4;

  auto dom = std::make_shared<LocalDomain>(Vec3(n, n, n), Vec3(0, 0, 0), 0);
  dom->set_radius(radius);
  // 8 physical fields + the 2 exchanged auxiliary divergence fields
  const char *names[10] = {"lnrho", "uux", "uuy", "uuz", "aax",
                           "aay",   "aaz", "ss",  "divu", "diva"};
  for (const char *nm : names) dom->add_data(sizeof(double), nm);
  dom->realize();

  ExchangeEngine eng({dom});
  // periodic self-exchange in two quantity groups: group 0 = the 8
  // fields before the div pass, group 1 = divu/diva before the main pass
  std::vector<int64_t> fields = {0, 1, 2, 3, 4, 5, 6, 7}, divs = {8, 9};
  for (int dz = -1; dz <= 1; ++dz)
    for (int dy = -1; dy <= 1; ++dy)
      for (int dx = -1; dx <= 1; ++dx) {
        const Vec3 d(dx, dy, dz);
        if (d == Vec3(0, 0, 0) || radius.dir(-d) == 0) continue;
        const Vec3 ext = LocalDomain::halo_extent(-d, dom->size(), radius);
        eng.add_translate(0, 0, dom->halo_pos(d, false), dom->halo_pos(-d, true), ext, 0, fields);
        eng.add_translate(0, 0, dom->halo_pos(d, false), dom->halo_pos(-d, true), ext, 1, divs);
      }
  eng.finalize();

  const Rect3 r = dom->compute_region();
  MhdCoeffs cf; // astaroth.conf-style spacing (models/astaroth.py DEFAULT_CONF)
  cf.dsx = cf.dsy = cf.dsz = 0.04908738521; // 2*pi/128
  for (int qi = 0; qi < 8; ++qi) {
    const double amp = qi == 0 ? 0.01 : 1e-3;
    init_harmonic_f64(eng, 0, qi, r, 0.0, amp, 2 * M_PI * (1 + qi % 3) / n,
                      2 * M_PI * ((qi / 3) % 3) / n, 2 * M_PI * (qi % 2) / n, 0.1 * qi, false);
    init_harmonic_f64(eng, 0, qi, r, 0.0, 0.0, 0, 0, 0, 0, true);
  }
  eng.sync_compute();

  const auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it) {
    for (int s = 0; s < 3; ++s) {
      eng.launch_translates(0); // X1: field halos
      eng.sync_translates();
      mhd_div_pass(eng, 0, r, cf);
      eng.sync_compute();
      eng.launch_translates(1); // X2: div halos only
      eng.sync_translates();
      mhd_substep(eng, 0, r, s, dt, cf); // scalar || momentum, two streams
      eng.sync_compute();
      dom->swap();
    }
  }
  const std::chrono::duration<double> el = std::chrono::steady_clock::now() - t0;

  const FieldStats st = field_stats(eng, 0, 1, r);
  printf("astaroth_native: %lld^3, %d iters, %.3f ms/iter, %.1f Mcell/s, uux rms %.3e\n",
         (long long)n, iters, el.count() / iters * 1e3, (double)n * n * n * iters / el.count() / 1e6,
         st.rms);
  if (!std::isfinite(st.rms)) return 1;
  return 0;
}
