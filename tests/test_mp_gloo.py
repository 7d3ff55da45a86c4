"""Multi-process exchange tests over gloo (world_size 2, CPU): the full
distributed planning + packed wire transport, verified with the ripple
full-region check. This is the CPU stand-in for the reference's
test_cuda_mpi tier (mpiexec -n 2)."""
import multiprocessing as mp
import os

import numpy as np
import pytest


def _worker(rank, world, port, q, radius_spec, n_local, size):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        import stencil_amd as sa
        from stencil_amd import _C

        import sys

        sys.path.insert(0, os.path.dirname(__file__))
        from util import check_full_regions, fill_interiors

        dd = sa.DistributedDomain(*size, backend="torch")
        if isinstance(radius_spec, int):
            dd.set_radius(radius_spec)
        else:
            r = _C.Radius.constant(radius_spec[0])
            for (d, v) in radius_spec[1]:
                r.set_dir(*d, v)
            dd.set_radius(r)
        dd.set_gpus([0] * n_local)
        h = dd.add_data(np.float32, "q")
        dd.realize()
        fill_interiors(dd, h)
        dd.exchange()
        check_full_regions(dd, h)
        # second exchange after swap exercises buffer reuse
        dd.swap()
        fill_interiors(dd, h, scale=2.0)
        dd.exchange()
        check_full_regions(dd, h, scale=2.0)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception as e:  # pragma: no cover
        import traceback

        q.put((rank, f"FAIL: {e}\n{traceback.format_exc()}"))


def _run_world(world, radius_spec, n_local=1, size=(12, 10, 8), port=29613):
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [
        ctx.Process(target=_worker, args=(r, world, port, q, radius_spec, n_local, size))
        for r in range(world)
    ]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in procs]
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def test_two_ranks_r1():
    _run_world(2, 1, port=29613)


def test_two_ranks_r2_multidomain():
    _run_world(2, 2, n_local=2, port=29617)


def test_two_ranks_asymmetric():
    _run_world(2, (1, [((1, 0, 0), 2)]), port=29621)


def test_four_ranks_r1():
    _run_world(4, 1, size=(16, 12, 10), port=29625)


def _machine_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        from stencil_amd.parallel.comm import Comm
        from stencil_amd.parallel.machine import Machine

        comm = Comm()
        m = Machine.build(comm, [rank])  # each rank "owns" one device slot
        assert m.num_nodes() == 1  # both ranks share the host
        assert m.classify(0, 1) == "colocated"
        assert m.classify(rank, rank) == "self"
        # both ranks see the same global inventory
        inv = [(g.host, g.pci, tuple(g.ranks)) for g in m.gpus]
        invs = comm.allgather_object(inv)
        assert invs[0] == invs[1]
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))


def test_machine_world2():
    """Machine inventory is identical and consistent across ranks"""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    port = 29571
    procs = [ctx.Process(target=_machine_worker, args=(r, 2, port, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=120) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _fuzz_worker(rank, world, port, q, seed):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import random

        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys

        sys.path.insert(0, os.path.dirname(__file__))
        import stencil_amd as sa
        from stencil_amd import _C
        from util import check_full_regions, fill_interiors

        rng = random.Random(seed)  # same seed -> same config on all ranks
        for case in range(3):
            size = tuple(rng.randint(8, 20) for _ in range(3))
            r = _C.Radius.constant(rng.randint(1, 2))
            if rng.random() < 0.5:  # sprinkle asymmetry
                r.set_dir(1, 0, 0, rng.randint(1, 3))
            n_local = rng.choice([1, 2])
            dd = sa.DistributedDomain(*size, backend="torch")
            dd.set_radius(r)
            dd.set_gpus([0] * n_local)
            hs = [dd.add_data(np.float32, f"q{i}") for i in range(rng.randint(1, 3))]
            dd.realize()
            for h in hs:
                fill_interiors(dd, h, scale=1.0 + h.index)
            for _ in range(rng.randint(1, 3)):
                dd.exchange()
                for h in hs:
                    check_full_regions(dd, h, scale=1.0 + h.index)
                dd.swap()
                for h in hs:
                    fill_interiors(dd, h, scale=1.0 + h.index)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))


@pytest.mark.parametrize("seed,port", [(7, 29581), (23, 29585)])
def test_fuzz_mp_exchange(seed, port):
    """randomized multi-rank exchange campaigns over gloo: sizes, radii
    (incl. asymmetric), quantity counts and subdomain counts vary; the
    ripple full-region check validates every cell incl. the injective
    per-pair wire tags under multiple transfers per rank pair"""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_fuzz_worker, args=(r, 2, port, q, seed)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _two_node_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys

        sys.path.insert(0, os.path.dirname(__file__))
        import stencil_amd as sa
        from stencil_amd.parallel import comm as comm_mod
        from util import check_full_regions, fill_interiors

        # simulate one rank per NODE: distinct hostnames drive the
        # two-level NodePartition and the non-colocated wire path (no IPC
        # possible across "nodes")
        orig_init = comm_mod.Comm.__init__

        def patched(self):
            orig_init(self)
            self.hostname = f"fakenode{rank}"

        comm_mod.Comm.__init__ = patched

        dd = sa.DistributedDomain(16, 12, 10, backend="torch")
        dd.set_radius(2)
        dd.set_gpus([0])
        h = dd.add_data(np.float32, "q")
        dd.realize()
        assert dd.comm.node_of_rank() == [0, 1]
        assert dd.comm.colocated_ranks() == [rank]
        assert dd.placement.part.sys_dim().tuple() != (1, 1, 1)  # node-level split
        assert dd.bytes_by_method["rccl"] > 0  # cross-node halos on the wire
        assert dd.bytes_by_method["ipc_kernel"] == 0
        fill_interiors(dd, h)
        dd.exchange()
        check_full_regions(dd, h)
        dd.swap()
        fill_interiors(dd, h, scale=2.0)
        dd.exchange()
        check_full_regions(dd, h, scale=2.0)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))


def test_two_node_simulation():
    """multi-NODE path (distinct hostnames): two-level partition, no
    colocated ranks, every cross-rank halo over the packed wire — the
    shape multi-node RCCL runs take, minus the fabric"""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_two_node_worker, args=(r, 2, 29591, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _hetero_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys

        sys.path.insert(0, os.path.dirname(__file__))
        import stencil_amd as sa
        from stencil_amd.parallel import comm as comm_mod
        from util import check_full_regions, fill_interiors

        # node A holds ranks 0,1; node B holds rank 2 -> heterogeneous
        # per-node GPU counts -> flat single-level partition fallback
        node = "nodeA" if rank < 2 else "nodeB"
        orig_init = comm_mod.Comm.__init__

        def patched(self):
            orig_init(self)
            self.hostname = node

        comm_mod.Comm.__init__ = patched

        dd = sa.DistributedDomain(18, 12, 10, backend="torch")
        dd.set_radius(1)
        dd.set_gpus([0])
        h = dd.add_data(np.float32, "q")
        dd.realize()
        assert dd.placement.uniform_nodes is False
        d = dd.placement.dim()
        assert d[0] * d[1] * d[2] == 3
        fill_interiors(dd, h)
        dd.exchange()
        check_full_regions(dd, h)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))


def test_heterogeneous_nodes_exchange():
    """heterogeneous per-node GPU counts (2+1) end-to-end: the flat
    placement fallback plans and exchanges correctly across 3 ranks"""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_hetero_worker, args=(r, 3, 29595, q)) for r in range(3)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(3)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"


def _methods_worker(rank, world, port, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        os.environ["MASTER_PORT"] = str(port)
        import torch.distributed as dist

        dist.init_process_group("gloo", rank=rank, world_size=world)
        import sys

        sys.path.insert(0, os.path.dirname(__file__))
        import stencil_amd as sa
        from util import check_full_regions, fill_interiors

        dd = sa.DistributedDomain(14, 10, 8, backend="torch")
        dd.set_radius(1)
        dd.set_gpus([0])
        # disable the colocated IPC transport via the Method API: halos
        # between colocated ranks must fall back to the packed wire
        dd.set_methods(sa.Method.DIRECT_KERNEL | sa.Method.RCCL)
        h = dd.add_data(np.float32, "q")
        dd.realize()
        assert dd.bytes_by_method["ipc_kernel"] == 0
        assert dd.bytes_by_method["rccl"] > 0
        fill_interiors(dd, h)
        dd.exchange()
        check_full_regions(dd, h)
        dist.destroy_process_group()
        q.put((rank, "ok"))
    except Exception:  # pragma: no cover
        import traceback

        q.put((rank, traceback.format_exc()))


def test_method_flags_disable_ipc():
    """Method API fallback chain (reference method.hpp first-match
    dispatch): with IPC_KERNEL cleared, colocated halos ride the wire"""
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    procs = [ctx.Process(target=_methods_worker, args=(r, 2, 29599, q)) for r in range(2)]
    for p in procs:
        p.start()
    results = [q.get(timeout=300) for _ in range(2)]
    for p in procs:
        p.join(timeout=60)
    for rank, status in results:
        assert status == "ok", f"rank {rank}: {status}"
