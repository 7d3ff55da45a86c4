// App compute ops (launched on a domain's compute stream).
#pragma once

#include "stencil_amd/core.hpp"
#include "stencil_amd/engine.hpp"

namespace stencil_amd {

// One Jacobi 7-point step of quantity qi over `region` (global coords):
// next = avg of 6 face neighbors of curr, with the reference's hot/cold
// sphere sources fixed inside `computeRegion` (bin/jacobi3d.cu:40-85).
// extendVec enables the pure-vector + LDS-rows fast path (1 = free
// extension, 2 = strict/no-extension for IPC ranks) by
// extending the row to aligned bounds: ONLY valid when every extended
// cell (x-halo cells, pitch-slack bytes, or exterior-shell cells that a
// LATER kernel on the SAME stream rewrites) is discard-safe and the
// domain has radius >= 1 -- the caller asserts this (the overlap
// interior and full-rect launches qualify).
void jacobi_step(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                 const Rect3 &computeRegion, int streamId = 0, int extendVec = 0);

// fill an fp32 region with `value` (curr or next buffer)
void fill_f32(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, float value,
              bool nextBuf);

// Whole-step hipGraph for the single-process single-domain jacobi path:
// [translate copy_batch -> full-region jacobi -> device-side table swap]
// captured once per buffer parity, replayed per step (~5 us host cost vs
// ~0.25 ms of per-step orchestration). launch() enqueues nSteps replays
// (alternating parities, host mirrors flipped); sync() blocks on them.
int64_t jacobi_graph_create(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                            const Rect3 &computeRegion);
// overlap variant: translate forked concurrent with the interior kernel
// inside the graph (STENCIL_AMD_GRAPH_OVERLAP=1; see csrc/src/jacobi.hip)
int64_t jacobi_graph_create_overlap(ExchangeEngine &eng, int dom, int64_t qi,
                                    const Rect3 &interior, const Rect3 &computeRegion,
                                    const std::vector<Rect3> &exteriors);
void jacobi_graph_launch(int64_t handle, int64_t nSteps);
void jacobi_graph_sync(int64_t handle);

// Multi-rank whole-step graphs (see csrc/src/jacobi.hip): per parity,
// A = [interior + translates + staged packs], B = [staged unpacks +
// exterior + device swap]; the caller runs the cross-rank barrier in
// between (RcclWire::barrier on jacobi_mr_graph_stream, or a host
// barrier with stream syncs around it).
int64_t jacobi_mr_graph_create(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &interior,
                               const Rect3 &computeRegion, const std::vector<Rect3> &exteriors,
                               int extendVec);
uintptr_t jacobi_mr_graph_stream(int64_t handle);
void jacobi_mr_graph_pre(int64_t handle);
void jacobi_mr_graph_post(int64_t handle);
void jacobi_mr_graph_sync(int64_t handle);

// physical coefficients of the MHD solver (see csrc/src/mhd.hip)
struct MhdCoeffs {
  double dsx = 1.0, dsy = 1.0, dsz = 1.0;
  double cs2 = 1.0;
  double cp_inv = 1.0;
  double nu = 5e-3, eta = 5e-3, chi = 5e-4;
};

// one RK3 substep (step in 0..2) of the 8-field 6th-order MHD system over
// `region` (global coords); reads curr, updates next in place (Williamson
// two-buffer form) -- caller swaps after each substep
// Separable-derivative MHD: the domain carries 10 fp64 quantities
// (8 physics fields + div u + div A). Per substep the app runs
//   exchange() -> mhd_div_pass -> exchange() -> mhd_substep
// so grad(div .) reduces to first derivatives of the exchanged div fields.
void mhd_div_pass(ExchangeEngine &eng, int dom, const Rect3 &region, const MhdCoeffs &cf,
                  int streamId = 0);
// Whole-substep hipGraphs for the world=1 single-domain astaroth shape:
// per RK3 substep and buffer parity, [X1 translates -> div -> X2
// translates -> scalar || momentum -> table swap] captured once;
// mhd_graph_iter replays 3*nIters substep graphs (host mirrors flipped).
int64_t mhd_graph_create(ExchangeEngine &eng, int dom, const Rect3 &region, double dt,
                         const MhdCoeffs &cf);
void mhd_graph_iter(int64_t handle, int64_t nIters = 1);
void mhd_graph_sync(int64_t handle);

// Multi-rank substep graphs (see csrc/src/mhd.hip): per substep the
// caller runs phase1 / barrier / phase2 / barrier / phase3 (barrier =
// RcclWire::barrier on mhd_mr_graph_stream, or a host barrier with a
// stream sync before it).
int64_t mhd_mr_graph_create(ExchangeEngine &eng, int dom, const Rect3 &interior,
                            const std::vector<Rect3> &exteriors, double dt, const MhdCoeffs &cf);
uintptr_t mhd_mr_graph_stream(int64_t handle);
void mhd_mr_phase1(int64_t handle);
void mhd_mr_phase2(int64_t handle);
void mhd_mr_phase3(int64_t handle);
void mhd_mr_graph_sync(int64_t handle);

void mhd_substep(ExchangeEngine &eng, int dom, const Rect3 &region, int step, double dt,
                 const MhdCoeffs &cf, int streamId = 0);

// fill an fp64 region with base + amp*sin(kx*x + ky*y + kz*z + phase)
// (deterministic smooth initial conditions, reproducible in NumPy)
void init_harmonic_f64(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, double base,
                       double amp, double kx, double ky, double kz, double phase, bool nextBuf);
// radial gaussian bump around (cx,cy,cz) (reference astaroth.cu
// radial_explosion_init_kernel): base + amp*exp(-r^2/(2 sigma^2))
void init_radial_f64(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, double base,
                     double amp, double cx, double cy, double cz, double sigma, bool nextBuf);

// min/max/RMS of a quantity over a region (reference: reductions.cuh)
struct FieldStats {
  double min, max, rms;
};
FieldStats field_stats(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                       bool nextBuf = false);

} // namespace stencil_amd
