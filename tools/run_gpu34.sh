#!/bin/bash
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu34.log
: > $LOG
echo "=== MHD suite with default ychunk ===" >> $LOG
timeout 500 python -m pytest tests/test_gpu_mhd.py tests/test_gpu_mp_mhd.py -x -q >> $LOG 2>&1
echo "=== astaroth default-config finals (ychunk=32 default) ===" >> $LOG
timeout 150 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 8 --warmup 2 2>&1 | grep astaroth, >> $LOG
timeout 250 python benchmarks/astaroth.py --gpus 1 --per-gpu 512 --iters 4 --warmup 1 2>&1 | grep astaroth, >> $LOG
timeout 250 python benchmarks/astaroth.py --gpus 1 --per-gpu 640 --iters 3 --warmup 1 2>&1 | grep astaroth, >> $LOG
echo "=== pipelined astaroth final ===" >> $LOG
timeout 250 python - <<'PYEOF' >> $LOG 2>&1
import sys, time
sys.path.insert(0, "/root/repo")
from stencil_amd.models.astaroth import Astaroth
app = Astaroth((256, 256, 256), gpus=[0]); app.realize(); app.init_fields()
for _ in range(2): app.step(overlap=False)
t0 = time.perf_counter(); app.run(10); dt = (time.perf_counter() - t0) / 10
print(f"astaroth pipelined: {dt*1e3:.3f} ms/iter, {256**3/dt/1e6:.0f} Mcell/s")
PYEOF
cat $LOG
