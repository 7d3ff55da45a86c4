#!/usr/bin/env python3
"""Build the stencil_amd native extension in-tree with hipcc for gfx950.

The extension is pure pybind11 + HIP (no libtorch link), so it imports on
CPU-only boxes (GPU calls fail only when actually made). Built .so lands in
stencil_amd/ so the gpurun snapshot carries it.
"""
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
CSRC = REPO / "csrc"
BUILD = REPO / "build"
ARCH = os.environ.get("STENCIL_AMD_ARCH", "gfx950")
HIPCC = os.environ.get("HIPCC", "/opt/rocm/bin/hipcc")


def pybind11_includes():
    import pybind11

    return [pybind11.get_include(), sysconfig.get_paths()["include"]]


def sources():
    return sorted(list((CSRC / "src").glob("*.cpp")) + list((CSRC / "src").glob("*.hip")))


def newest_header_mtime():
    hs = list((CSRC / "include").rglob("*.hpp"))
    return max(h.stat().st_mtime for h in hs) if hs else 0.0


def build(verbose=True):
    BUILD.mkdir(exist_ok=True)
    inc = [str(CSRC / "include")] + pybind11_includes()
    cflags = [
        "-O3",
        "-std=c++17",
        "-fPIC",
        f"--offload-arch={ARCH}",
        "-Wall",
        "-Wno-unused-function",
    ] + [f"-I{i}" for i in inc]

    hmtime = newest_header_mtime()
    objs = []
    for src in sources():
        obj = BUILD / (src.stem + ".o")
        objs.append(obj)
        if obj.exists() and obj.stat().st_mtime > max(src.stat().st_mtime, hmtime):
            continue
        cmd = [HIPCC, "-c", str(src), "-o", str(obj)] + cflags
        if src.suffix == ".cpp":
            cmd.insert(1, "-x")
            cmd.insert(2, "hip")  # host+device aware; harmless for host-only TUs
        if verbose:
            print("[build]", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)

    suffix = sysconfig.get_config_var("EXT_SUFFIX")
    out = REPO / "stencil_amd" / f"_C{suffix}"
    if not out.exists() or any(o.stat().st_mtime > out.stat().st_mtime for o in objs):
        cmd = [HIPCC, "-shared", "-fPIC", "-o", str(out)] + [str(o) for o in objs] + [
            "-L/opt/rocm/lib",
            "-lroctx64",
            "-lrccl",
        ]
        if verbose:
            print("[link]", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)

    # native C++ example binaries (no Python dependency)
    core_objs = [o for o in objs if o.stem != "bindings"]
    for ex in sorted((REPO / "examples").glob("*.cpp")):
        exe = BUILD / ex.stem
        if exe.exists() and exe.stat().st_mtime > max(
            [ex.stat().st_mtime, hmtime] + [o.stat().st_mtime for o in core_objs]
        ):
            continue
        exo = BUILD / (ex.stem + "_main.o")
        cmd = [HIPCC, "-x", "hip", "-c", str(ex), "-o", str(exo)] + cflags
        if verbose:
            print("[exe-compile]", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)
        cmd = [HIPCC, str(exo)] + [str(o) for o in core_objs] + [
            "-L/opt/rocm/lib",
            "-lroctx64",
            "-lrccl",
            "-o",
            str(exe),
        ]
        if verbose:
            print("[exe-link]", " ".join(cmd), flush=True)
        subprocess.check_call(cmd)
    return out


if __name__ == "__main__":
    build()
    print("built OK")
