// Quadratic Assignment Problem solver for topology-aware placement.
//
// MI355X-native equivalent of the reference's qap.hpp (reference:
// include/stencil/qap.hpp:24-180). Given a communication-weight matrix w
// (subdomain pair -> bytes) and a distance matrix d (GPU pair -> 1/bandwidth),
// find an assignment f of subdomains to GPUs minimizing
//   cost(f) = sum_{a,b} w(a,b) * d(f(a), f(b)).
//
// On a single MI355X node all 8 GPUs are one xGMI hop apart, so the matrix
// is near-uniform and any assignment is near-optimal; the solver still
// matters for multi-node placement and for heterogeneous link topologies.
//
// Two strategies, like the reference: exact branch-free next_permutation
// search with a wall-clock cap, and a 2-swap hill climb with incremental
// cost evaluation for larger n.
#pragma once

#include <algorithm>
#include <chrono>
#include <cstdint>
#include <numeric>
#include <vector>

namespace stencil_amd {

// dense row-major square matrix helper
struct SqMat {
  int64_t n = 0;
  std::vector<double> v;
  SqMat() = default;
  explicit SqMat(int64_t n_, double fill = 0.0) : n(n_), v(n_ * n_, fill) {}
  double &at(int64_t i, int64_t j) { return v[i * n + j]; }
  double at(int64_t i, int64_t j) const { return v[i * n + j]; }
};

namespace qap {

inline double cost(const SqMat &w, const SqMat &d, const std::vector<int64_t> &f) {
  double c = 0;
  for (int64_t a = 0; a < w.n; ++a)
    for (int64_t b = 0; b < w.n; ++b)
      c += w.at(a, b) * d.at(f[a], f[b]);
  return c;
}

// Exact search over all permutations, with a time cap (seconds). For n <= 8
// (one node of MI355X GPUs) this is at most 40320 permutations.
inline std::vector<int64_t> solve_exact(const SqMat &w, const SqMat &d, double timeoutSec = 10.0) {
  std::vector<int64_t> f(w.n), best(w.n);
  std::iota(f.begin(), f.end(), 0);
  best = f;
  double bestCost = cost(w, d, f);
  const auto start = std::chrono::steady_clock::now();
  int check = 0;
  while (std::next_permutation(f.begin(), f.end())) {
    const double c = cost(w, d, f);
    if (c < bestCost) {
      bestCost = c;
      best = f;
    }
    if (++check % 512 == 0) {
      const std::chrono::duration<double> e = std::chrono::steady_clock::now() - start;
      if (e.count() > timeoutSec) break;
    }
  }
  return best;
}

// 2-swap hill climb with incremental cost delta; restarts until no
// improving swap exists.
inline std::vector<int64_t> solve_climb(const SqMat &w, const SqMat &d) {
  const int64_t n = w.n;
  std::vector<int64_t> f(n);
  std::iota(f.begin(), f.end(), 0);

  // delta of swapping f[i] and f[j]
  auto swap_delta = [&](int64_t i, int64_t j) {
    double delta = 0;
    const int64_t fi = f[i], fj = f[j];
    for (int64_t k = 0; k < n; ++k) {
      if (k == i || k == j) continue;
      const int64_t fk = f[k];
      delta += w.at(i, k) * (d.at(fj, fk) - d.at(fi, fk));
      delta += w.at(k, i) * (d.at(fk, fj) - d.at(fk, fi));
      delta += w.at(j, k) * (d.at(fi, fk) - d.at(fj, fk));
      delta += w.at(k, j) * (d.at(fk, fi) - d.at(fk, fj));
    }
    delta += w.at(i, j) * (d.at(fj, fi) - d.at(fi, fj));
    delta += w.at(j, i) * (d.at(fi, fj) - d.at(fj, fi));
    return delta;
  };

  bool improved = true;
  while (improved) {
    improved = false;
    for (int64_t i = 0; i < n; ++i)
      for (int64_t j = i + 1; j < n; ++j)
        if (swap_delta(i, j) < -1e-12) {
          std::swap(f[i], f[j]);
          improved = true;
        }
  }
  return f;
}

// Entry point: exact for small n, hill-climb beyond.
inline std::vector<int64_t> solve(const SqMat &w, const SqMat &d, double timeoutSec = 10.0) {
  if (w.n <= 9) return solve_exact(w, d, timeoutSec);
  return solve_climb(w, d);
}

} // namespace qap
} // namespace stencil_amd
