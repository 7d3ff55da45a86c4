#!/bin/bash
# Sanitizer lane (the reference wired cuda-memcheck into ctest,
# test/CMakeLists.txt:34,49; ROCm's equivalent serialization knobs make
# every kernel launch and copy synchronous so asynchronous-race bugs
# surface as immediate failures at the faulting call site):
#   AMD_SERIALIZE_KERNEL=3  sync before AND after every kernel launch
#   AMD_SERIALIZE_COPY=3    sync before AND after every copy
#   HIP_LAUNCH_BLOCKING=1   belt-and-suspenders host-blocking launches
# Usage: tools/run_sanitized.sh [pytest args...]   (defaults to -m gpu)
set -e
cd "$(dirname "$0")/.."
export AMD_SERIALIZE_KERNEL=3
export AMD_SERIALIZE_COPY=3
export HIP_LAUNCH_BLOCKING=1
args=("$@")
if [ ${#args[@]} -eq 0 ]; then args=(-m gpu); fi
exec python -m pytest tests -x -q "${args[@]}"
