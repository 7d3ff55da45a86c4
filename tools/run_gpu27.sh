#!/bin/bash
# Final round-1 validation: full GPU test suite, smoke, headline benches,
# soaks, multi-rank smoke. ~12 min.
set -x
cd /root/repo
export PYTHONPATH=/root/repo
LOG=/root/repo/gpurun_out/gpu27.log
: > $LOG
echo "=== full GPU test suite ===" >> $LOG
timeout 900 python -m pytest tests/ -m gpu -q >> $LOG 2>&1
echo "=== smoke() ===" >> $LOG
timeout 300 python -c "import __graft_entry__; __graft_entry__.smoke()" >> $LOG 2>&1
echo "=== bench.py final (default flags) ===" >> $LOG
timeout 200 python bench.py 2>&1 | grep -E '^\{' >> $LOG
echo "=== eager-path bench (address-aligned head fix) ===" >> $LOG
STENCIL_AMD_STEP_GRAPH=0 timeout 200 python bench.py --gpus 1 --steps 20 --warmup 3 2>&1 | grep -E '^\{' >> $LOG
echo "=== astaroth final ===" >> $LOG
timeout 200 python benchmarks/astaroth.py --gpus 1 --per-gpu 256 --iters 10 --warmup 2 2>&1 | grep astaroth, >> $LOG
echo "=== soak: jacobi graph 500 iters + finite check ===" >> $LOG
timeout 900 python - <<'PYEOF' >> $LOG 2>&1
import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
from stencil_amd.models.jacobi3d import Jacobi3D
app = Jacobi3D((512, 512, 512), backend="native", gpus=[0]); app.realize()
t0 = time.perf_counter(); app.run(500); dt = time.perf_counter() - t0
lo, hi = app.dd.local_rect(0)
a = app.dd.read_global(0, lo, hi, app.h)
assert np.isfinite(a).all() and a.max() <= 1.0 and a.min() >= 0.0 and a.max() > 0.9
print(f"jacobi soak 500 iters @512^3 OK: {dt/500*1e3:.3f} ms/iter, range [{a.min():.3f},{a.max():.3f}]")
PYEOF
echo "=== soak: astaroth graph 100 iters + finite check ===" >> $LOG
timeout 900 python - <<'PYEOF' >> $LOG 2>&1
import sys, time
sys.path.insert(0, "/root/repo")
import numpy as np
from stencil_amd.models.astaroth import Astaroth, FIELDS
app = Astaroth((256, 256, 256), gpus=[0]); app.realize(); app.init_fields()
t0 = time.perf_counter(); app.run(100); dt = time.perf_counter() - t0
for n in FIELDS:
    a = app.read_field(0, n)
    assert np.isfinite(a).all(), n
print(f"astaroth soak 100 iters @256^3 OK: {dt/100*1e3:.3f} ms/iter")
PYEOF
echo "=== 4-rank bench.py smoke (new barrier path, gloo fallback) ===" >> $LOG
timeout 400 python -m torch.distributed.run --nnodes=1 --nproc-per-node 4 --master-addr 127.0.0.1 --master-port 29793 bench.py --gpus 4 --per-gpu 192 --steps 8 --warmup 2 2>&1 | grep -E '^\{' >> $LOG
echo "=== 2500^3 graph-mode jacobi (big memory) ===" >> $LOG
timeout 400 python bench.py --gpus 1 --per-gpu 2500 --steps 5 --warmup 1 2>&1 | grep -E '^\{' >> $LOG
tail -35 $LOG
