"""Multi-rank MHD correctness (2 ranks on one GPU over the IPC transport):
the separable-derivative flow runs TWO exchanges per substep; results must
match the NumPy global-periodic reference exactly like the single-process
case. This is the single-box analog of the driver's 8-GPU run."""
import multiprocessing as mp
import os

import numpy as np
import pytest

pytestmark = pytest.mark.gpu


def _worker(rank, world, port, q, step_graph=True):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        os.environ["STENCIL_AMD_WIRE"] = "cpu"
        os.environ["STENCIL_AMD_STEP_GRAPH"] = "1" if step_graph else "0"
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys

        sys.path.insert(0, os.path.dirname(__file__))
        from stencil_amd.models import mhd_ref as M
        from stencil_amd.models.astaroth import Astaroth, FIELDS
        from test_gpu_mhd import global_init

        size = (24, 24, 24)
        app = Astaroth(size, backend="native")
        app.realize()
        # the multi-rank substep-graph path must engage iff graphs are on
        # (every cross-rank halo here is IPC)
        assert (app._mr_graph is not None) == step_graph
        app.init_fields()

        cf = {
            k: float(app.conf[k])
            for k in ("dsx", "dsy", "dsz", "cs2", "cp_inv", "nu", "eta", "chi")
        }
        curr = global_init(size)
        nxt = [np.zeros(curr[0].shape) for _ in range(8)]
        dt = 1e-4
        for _ in range(2):
            app.step(dt=dt)  # multi-rank -> overlap path with 2 exchanges
            for s in range(3):
                curr, nxt = M.substep(curr, nxt, s, dt, cf)
        for li in range(app.dd.num_local()):
            lo, hi = app.dd.local_rect(li)
            for qi, name in enumerate(FIELDS):
                got = app.read_field(li, name)
                want = curr[qi][lo[2] : hi[2], lo[1] : hi[1], lo[0] : hi[0]]
                np.testing.assert_allclose(
                    got, want, rtol=1e-9, atol=1e-12, err_msg=f"{name} li={li} rank={rank}"
                )
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


@pytest.mark.parametrize("step_graph,port", [(True, 29771), (False, 29775)])
def test_mhd_two_ranks_matches_numpy(step_graph, port):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_worker, args=(r, 2, port, q, step_graph)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in procs]
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"
