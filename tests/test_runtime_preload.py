"""Regression test for the dual-HIP-runtime footgun: importing
stencil_amd before torch must still leave exactly ONE libamdhip64 in the
process (stencil_amd/__init__.py preloads torch's bundled runtime).
With two runtimes, the kernel driver registers the process once and the
loser's hipGetDeviceCount sees 0 GPUs (observed under torchrun)."""
import subprocess
import sys

SCRIPT = r"""
import stencil_amd   # must come FIRST (the failing order)
import torch
libs = sorted(set(l.split()[-1] for l in open("/proc/self/maps")
                  if "libamdhip64" in l))
print(len(libs), libs)
assert len(libs) == 1, f"dual HIP runtime loaded: {libs}"
"""


def test_single_hip_runtime_package_first():
    r = subprocess.run([sys.executable, "-c", SCRIPT], capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
