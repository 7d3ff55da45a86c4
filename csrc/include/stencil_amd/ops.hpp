// App compute ops (launched on a domain's compute stream).
#pragma once

#include "stencil_amd/core.hpp"
#include "stencil_amd/engine.hpp"

namespace stencil_amd {

// One Jacobi 7-point step of quantity qi over `region` (global coords):
// next = avg of 6 face neighbors of curr, with the reference's hot/cold
// sphere sources fixed inside `computeRegion` (bin/jacobi3d.cu:40-85).
void jacobi_step(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region,
                 const Rect3 &computeRegion);

// fill an fp32 region with `value` (curr or next buffer)
void fill_f32(ExchangeEngine &eng, int dom, int64_t qi, const Rect3 &region, float value,
              bool nextBuf);

} // namespace stencil_amd
