#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu30.log
: > $LOG
echo "=== numerics: LDS graph vs eager vs torch (+ odd sizes) ===" >> $LOG
timeout 400 python -m pytest tests/test_gpu_native.py -x -q >> $LOG 2>&1
timeout 200 python - <<'PYEOF' >> $LOG 2>&1
import sys
sys.path.insert(0, "/root/repo"); sys.path.insert(0, "/root/repo/tests")
import numpy as np
from stencil_amd.models.jacobi3d import Jacobi3D
# odd shapes stress the clamp paths: extY%4 in {1,2,3}, extX%4 != 0
for size in [(37, 29, 22), (50, 51, 19), (24, 18, 14), (65, 43, 33)]:
    outs = []
    for env in ("1", "0"):
        import os
        os.environ["STENCIL_AMD_STEP_GRAPH"] = env
        app = Jacobi3D(size, backend="native", gpus=[0]); app.realize()
        for _ in range(3): app.step()
        lo, hi = app.dd.local_rect(0)
        outs.append(app.dd.read_global(0, lo, hi, app.h))
    assert np.array_equal(outs[0], outs[1]), f"LDS graph != eager at {size}"
    print("clamp-ok", size)
PYEOF
echo "=== bench A/B: LDS vs no-LDS graph vs eager (one box) ===" >> $LOG
for i in 1 2; do
  timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('graph+lds ', round(d['ms_per_step'],4), round(d['value']/1e9,1))" >> $LOG
  STENCIL_JAC_LDS=0 timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('graph     ', round(d['ms_per_step'],4), round(d['value']/1e9,1))" >> $LOG
done
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 1024 --iters 15 2>&1 | grep jacobi3d, >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 512 --iters 40 2>&1 | grep jacobi3d, >> $LOG
timeout 300 python benchmarks/jacobi3d.py --gpus 1 --strong --size 2048 --radius 2 --iters 8 2>&1 | grep jacobi3d, >> $LOG
cat $LOG
