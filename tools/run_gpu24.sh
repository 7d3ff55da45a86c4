#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu24.log
: > $LOG
echo "=== bench.py pipelined 750 (2 runs) + spin off control ===" >> $LOG
for i in 1 2; do
  timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('pipelined', d['ms_per_step'], d['value']/1e9)" >> $LOG
done
STENCIL_AMD_SPIN=0 timeout 200 python bench.py --gpus 1 --steps 30 --warmup 5 2>&1 | grep -E '^\{' | python -c "import json,sys; d=json.load(sys.stdin); print('nospin  ', d['ms_per_step'], d['value']/1e9)" >> $LOG
echo "=== per-step (jacobi3d.py benchmark, per-iter sync) for contrast ===" >> $LOG
timeout 200 python benchmarks/jacobi3d.py --gpus 1 --size 750 --iters 30 2>&1 | grep jacobi3d, >> $LOG
echo "=== astaroth pipelined: model.run via quick python ===" >> $LOG
timeout 250 python - <<'PYEOF' >> $LOG 2>&1
import time, sys
sys.path.insert(0, "/root/repo")
from stencil_amd.models.astaroth import Astaroth
app = Astaroth((256, 256, 256), gpus=[0]); app.realize(); app.init_fields()
for _ in range(2): app.step(overlap=False)  # warmup
t0 = time.perf_counter(); app.run(8); dt = (time.perf_counter() - t0) / 8
print(f"astaroth pipelined: {dt*1e3:.3f} ms/iter, {256**3/dt/1e6:.0f} Mcell/s")
t0 = time.perf_counter()
for _ in range(8): app.step(overlap=False)
dt = (time.perf_counter() - t0) / 8
print(f"astaroth per-step : {dt*1e3:.3f} ms/iter, {256**3/dt/1e6:.0f} Mcell/s")
PYEOF
echo "=== numerics: pipelined == per-step (jacobi) ===" >> $LOG
timeout 200 python - <<'PYEOF' >> $LOG 2>&1
import sys
sys.path.insert(0, "/root/repo")
sys.path.insert(0, "/root/repo/tests")
import numpy as np
from stencil_amd.models.jacobi3d import Jacobi3D
outs = []
for mode in ("run", "step"):
    app = Jacobi3D((64, 48, 40), backend="native", gpus=[0]); app.realize()
    if mode == "run": app.run(6)
    else:
        for _ in range(6): app.step()
    lo, hi = app.dd.local_rect(0)
    outs.append(app.dd.read_global(0, lo, hi, app.h))
assert np.array_equal(outs[0], outs[1]), "pipelined != per-step!"
print("pipelined == per-step OK")
PYEOF
cat $LOG
