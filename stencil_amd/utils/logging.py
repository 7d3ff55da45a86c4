"""Leveled stderr logging with rank + file:line prefix (reference:
include/stencil/logging.hpp). Level from STENCIL_LOG (spew|debug|info|
warn|error), default info."""
from __future__ import annotations

import inspect
import os
import sys

_LEVELS = {"spew": 0, "debug": 1, "info": 2, "warn": 3, "error": 4, "fatal": 5}
_LEVEL = _LEVELS.get(os.environ.get("STENCIL_LOG", "info").lower(), 2)


def _rank() -> str:
    return os.environ.get("RANK", "0")


def _emit(level: str, *args):
    if _LEVELS[level] < _LEVEL:
        return
    fr = inspect.stack()[2]
    loc = f"{os.path.basename(fr.filename)}:{fr.lineno}"
    print(f"[{level.upper()}] {loc} {{{_rank()}}} " + " ".join(str(a) for a in args),
          file=sys.stderr, flush=True)


def spew(*a):
    _emit("spew", *a)


def debug(*a):
    _emit("debug", *a)


def info(*a):
    _emit("info", *a)


def warn(*a):
    _emit("warn", *a)


def error(*a):
    _emit("error", *a)


def fatal(*a):
    _emit("fatal", *a)
    sys.exit(1)
