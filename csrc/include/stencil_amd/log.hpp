// Leveled stderr logging for the native (C++) layer.
//
// Reference parity: include/stencil/logging.hpp (compile-time-leveled
// LOG_* macros printing file:line {rank}). Here the level is runtime
// (STENCIL_LOG=spew|debug|info|warn|error, same env the Python
// utils/logging.py reads) and the rank comes from STENCIL_RANK/RANK.
#pragma once

#include <cstdio>
#include <cstdlib>
#include <cstring>

namespace stencil_amd {
namespace logging {

enum Level { kSpew = 0, kDebug, kInfo, kWarn, kError, kOff };

inline int level() {
  static int lv = -1;
  if (lv < 0) {
    const char *e = getenv("STENCIL_LOG");
    lv = kWarn;
    if (e) {
      if (!strcmp(e, "spew")) lv = kSpew;
      else if (!strcmp(e, "debug")) lv = kDebug;
      else if (!strcmp(e, "info")) lv = kInfo;
      else if (!strcmp(e, "warn")) lv = kWarn;
      else if (!strcmp(e, "error")) lv = kError;
      else if (!strcmp(e, "off")) lv = kOff;
    }
  }
  return lv;
}

inline int rank() {
  static int r = -2;
  if (r == -2) {
    const char *e = getenv("STENCIL_RANK");
    if (!e) e = getenv("RANK");
    r = e ? atoi(e) : -1;
  }
  return r;
}

} // namespace logging
} // namespace stencil_amd

#define STENCIL_LOG_AT(lvl, tag, ...)                                                              \
  do {                                                                                             \
    if ((lvl) >= ::stencil_amd::logging::level()) {                                                \
      fprintf(stderr, "[%s] %s:%d {%d} ", tag, __FILE__, __LINE__,                                 \
              ::stencil_amd::logging::rank());                                                     \
      fprintf(stderr, __VA_ARGS__);                                                                \
      fprintf(stderr, "\n");                                                                       \
    }                                                                                              \
  } while (0)

#define LOG_SPEW(...) STENCIL_LOG_AT(::stencil_amd::logging::kSpew, "SPEW", __VA_ARGS__)
#define LOG_DEBUG(...) STENCIL_LOG_AT(::stencil_amd::logging::kDebug, "DEBUG", __VA_ARGS__)
#define LOG_INFO(...) STENCIL_LOG_AT(::stencil_amd::logging::kInfo, "INFO", __VA_ARGS__)
#define LOG_WARN(...) STENCIL_LOG_AT(::stencil_amd::logging::kWarn, "WARN", __VA_ARGS__)
#define LOG_ERROR(...) STENCIL_LOG_AT(::stencil_amd::logging::kError, "ERROR", __VA_ARGS__)
