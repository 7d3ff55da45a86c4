// Pure-C++ Astaroth-class MHD driver on the C++ DistributedDomain — the
// orchestrator plans the quantity-group exchanges (8 physics fields vs 2
// divergence fields) exactly like the Python model (models/astaroth.py);
// no hand-wired translate jobs (reference astaroth/astaroth.cu is the
// C++-only equivalent app).
//
//   build/astaroth_native [edge=128] [iters=5] [ngpus=1]
#include <chrono>
#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <vector>

#include "stencil_amd/distributed.hpp"
#include "stencil_amd/ops.hpp"

using namespace stencil_amd;

int main(int argc, char **argv) {
  const int64_t n = argc > 1 ? atoll(argv[1]) : 128;
  const int iters = argc > 2 ? atoi(argv[2]) : 5;
  const int ngpus = argc > 3 ? atoi(argv[3]) : 1;
  const double dt = 1e-4;

  DistributedDomain dd(n, n, n);
  dd.set_radius(3);
  // 8 physical fields + the 2 exchanged auxiliary divergence fields
  const char *names[10] = {"lnrho", "uux", "uuy", "uuz", "aax",
                           "aay",   "aaz", "ss",  "divu", "diva"};
  for (const char *nm : names) dd.add_data<double>(nm);
  dd.set_exchange_groups({{0, 1, 2, 3, 4, 5, 6, 7}, {8, 9}});
  if (ngpus > 1) {
    std::vector<int> gpus;
    // STENCIL_FAKE_GPUS=1: put every subdomain on device 0 (the
    // same-GPU fake-multi-GPU test trick, reference test_exchange.cu:52)
    const char *fake = getenv("STENCIL_FAKE_GPUS");
    for (int i = 0; i < ngpus; ++i) gpus.push_back(fake && fake[0] == '1' ? 0 : i);
    dd.set_gpus(gpus);
  }
  dd.realize();

  MhdCoeffs cf; // astaroth.conf-style spacing (models/astaroth.py DEFAULT_CONF)
  cf.dsx = cf.dsy = cf.dsz = 0.04908738521; // 2*pi/128
  for (int li = 0; li < dd.num_local(); ++li) {
    const Rect3 r = dd.local_rect(li);
    for (int qi = 0; qi < 8; ++qi) {
      const double amp = qi == 0 ? 0.01 : 1e-3;
      init_harmonic_f64(dd.engine(), li, qi, r, 0.0, amp, 2 * M_PI * (1 + qi % 3) / n,
                        2 * M_PI * ((qi / 3) % 3) / n, 2 * M_PI * (qi % 2) / n, 0.1 * qi, false);
      init_harmonic_f64(dd.engine(), li, qi, r, 0.0, 0.0, 0, 0, 0, 0, true);
    }
  }
  dd.engine().sync_compute();

  const auto t0 = std::chrono::steady_clock::now();
  for (int it = 0; it < iters; ++it) {
    for (int s = 0; s < 3; ++s) {
      dd.exchange(0); // X1: field halos
      for (int li = 0; li < dd.num_local(); ++li)
        mhd_div_pass(dd.engine(), li, dd.local_rect(li), cf);
      dd.engine().sync_compute();
      dd.exchange(1); // X2: div halos only
      for (int li = 0; li < dd.num_local(); ++li)
        mhd_substep(dd.engine(), li, dd.local_rect(li), s, dt, cf); // scalar || momentum
      dd.engine().sync_compute();
      dd.swap();
    }
  }
  const std::chrono::duration<double> el = std::chrono::steady_clock::now() - t0;

  const FieldStats st = field_stats(dd.engine(), 0, 1, dd.local_rect(0));
  if (dd.rank() == 0)
    printf("astaroth_native: %lld^3 world=%d gpus=%d, %d iters, %.3f ms/iter, %.1f Mcell/s, "
           "uux rms %.3e\n",
           (long long)n, dd.world(), ngpus, iters, el.count() / iters * 1e3,
           (double)n * n * n * iters / el.count() / 1e6, st.rms);
  if (!std::isfinite(st.rms)) return 1;
  return 0;
}
