"""C++/Python orchestrator parity (no GPU): the C++ placement + planner
(csrc placement.hpp / planning.hpp, used by the C++ DistributedDomain) must
produce the SAME assignment, translate list, wire items, message order,
layout and tags as the Python orchestrator — a C++ rank and a Python rank
of one job must agree on the wire format."""
import pytest

from stencil_amd import _C
from stencil_amd.parallel.placement import (
    NodeAwarePlacement,
    Slot,
    TrivialPlacement,
)
from stencil_amd.parallel.planning import (
    pair_seq_tags,
    plan_exchange,
    wire_layout,
)


def make_slots(n_nodes, gpn):
    out = []
    rank = 0
    for node in range(n_nodes):
        for g in range(gpn):
            out.append(Slot(rank, 0, g, node))
            rank += 1
    return out


def py_placement(strategy, size, radius, slots):
    cls = {"trivial": TrivialPlacement, "node_aware": NodeAwarePlacement}[strategy]
    return cls(size, radius, slots)


CASES = [
    ((48, 40, 32), 1, 1, 4, "trivial"),
    ((48, 40, 32), 1, 1, 4, "node_aware"),
    ((64, 64, 64), 2, 2, 4, "node_aware"),
    ((30, 30, 30), 1, 1, 8, "node_aware"),
    ((33, 31, 29), 3, 1, 2, "trivial"),
    ((40, 36, 28), "asym", 1, 4, "node_aware"),
    ((40, 36, 28), "fec", 1, 8, "trivial"),
]


def make_radius(r):
    if r == "asym":  # the reference's +x=2/-x=1 asymmetric test shape
        rad = _C.Radius.constant(1)
        rad.set_dir(1, 0, 0, 2)
        rad.set_dir(-1, 0, 0, 1)
        rad.set_dir(1, 1, 0, 2)
        return rad
    if r == "fec":  # face/edge/corner radii 2/1/0
        return _C.Radius.face_edge_corner(2, 1, 0)
    return _C.Radius.constant(r)


@pytest.mark.parametrize("size,r,n_nodes,gpn,strategy", CASES)
def test_cpp_python_plan_parity(size, r, n_nodes, gpn, strategy):
    radius = make_radius(r)
    slots = make_slots(n_nodes, gpn)
    slot_tuples = [(s.rank, s.local_id, s.cuda, s.node) for s in slots]
    cpp = _C.cpp_plan(
        _C.Vec3(*size), radius, 0, slot_tuples, strategy
    )
    pp = py_placement(strategy, size, radius, slots)

    assert tuple(cpp["dim"]) == pp.dim()
    # identical subdomain -> (rank, local, cuda) assignment
    n = pp.dim()[0] * pp.dim()[1] * pp.dim()[2]
    for gid in range(n):
        idx = pp.dimensionize(gid)
        assert tuple(cpp["assign"][gid]) == (
            pp.get_rank(idx), pp.get_subdomain_id(idx), pp.get_cuda(idx)
        )

    for rank in range(len(slots)):
        cppr = _C.cpp_plan(_C.Vec3(*size), radius, rank, slot_tuples, strategy)
        plan = plan_exchange(pp, radius, rank)
        tags = pair_seq_tags(plan)
        assert len(cppr["translates"]) == len(plan.translates)
        for (sl, dl, d, e), t in zip(cppr["translates"], plan.translates):
            assert (sl, dl, tuple(d), tuple(e)) == (t.src_local, t.dst_local, t.dir, t.ext)
        for key, py_items in (("sends", plan.sends), ("recvs", plan.recvs)):
            cpp_items = cppr[key]
            assert len(cpp_items) == len(py_items)
            for (peer, sg, dg, li, msgs, tag), it in zip(cpp_items, py_items):
                assert (peer, sg, dg, li) == (it.peer_rank, it.src_gid, it.dst_gid, it.local_id)
                assert tag == tags[(it.peer_rank, it.src_gid, it.dst_gid)]
                assert len(msgs) == len(it.messages)
                for (d, msg, mdg, e), m in zip(msgs, it.messages):
                    assert (tuple(d), msg, mdg, tuple(e)) == (m.dir, m.src_gid, m.dst_gid, m.ext)


def test_cpp_wire_layout_parity():
    from stencil_amd.parallel.planning import Message

    msgs = [
        Message((1, 0, 0), 0, 1, (2, 5, 7)),
        Message((0, -1, 0), 0, 1, (16, 2, 7)),
        Message((1, 1, 0), 0, 1, (2, 2, 7)),
    ]
    elem_sizes = [4, 8, 2]
    for qis in ([0, 1, 2], [1], [0, 2]):
        total, chunks = wire_layout(msgs, elem_sizes, qis)
        ctotal, cchunks = _C.cpp_wire_layout(
            [((m.dir), m.src_gid, m.dst_gid, (m.ext)) for m in msgs], elem_sizes, qis
        )
        assert ctotal == total
        assert [tuple(c) for c in cchunks] == chunks


def test_cpp_python_plan_parity_fuzz():
    """randomized configs: sizes, radii (incl. asymmetric diagonals),
    node shapes, strategies — the C++ planner must agree with Python on
    every item, message, extent and tag"""
    import random

    rng = random.Random(42)
    for case in range(30):
        size = tuple(rng.randint(6, 40) for _ in range(3))
        radius = _C.Radius.constant(rng.randint(0, 3))
        for _ in range(rng.randint(0, 4)):
            d = (rng.randint(-1, 1), rng.randint(-1, 1), rng.randint(-1, 1))
            if d != (0, 0, 0):
                radius.set_dir(*d, rng.randint(0, 3))
        n_nodes = rng.choice([1, 1, 2])
        gpn = rng.choice([1, 2, 4, 8])
        strategy = rng.choice(["trivial", "node_aware"])
        slots = make_slots(n_nodes, gpn)
        slot_tuples = [(s.rank, s.local_id, s.cuda, s.node) for s in slots]
        pp = py_placement(strategy, size, radius, slots)
        for rank in range(min(len(slots), 3)):
            cpp = _C.cpp_plan(_C.Vec3(*size), radius, rank, slot_tuples, strategy)
            plan = plan_exchange(pp, radius, rank)
            tags = pair_seq_tags(plan)
            ctx = f"case {case} size={size} gpn={gpn} {strategy} rank={rank}"
            assert len(cpp["translates"]) == len(plan.translates), ctx
            for (sl, dl, d, e), t in zip(cpp["translates"], plan.translates):
                assert (sl, dl, tuple(d), tuple(e)) == (
                    t.src_local, t.dst_local, t.dir, t.ext), ctx
            for key, py_items in (("sends", plan.sends), ("recvs", plan.recvs)):
                assert len(cpp[key]) == len(py_items), ctx
                for (peer, sg, dg, li, msgs, tag), it in zip(cpp[key], py_items):
                    assert (peer, sg, dg, li) == (
                        it.peer_rank, it.src_gid, it.dst_gid, it.local_id), ctx
                    assert tag == tags[(it.peer_rank, it.src_gid, it.dst_gid)], ctx
                    assert [(tuple(d), m1, m2, tuple(e)) for d, m1, m2, e in msgs] == [
                        (m.dir, m.src_gid, m.dst_gid, m.ext) for m in it.messages], ctx


def test_cpp_python_parity_heterogeneous_nodes():
    """uneven per-node GPU counts: both planners fall back to the flat
    single-level partition and agree exactly"""
    size = (30, 24, 18)
    radius = _C.Radius.constant(1)
    slots = make_slots(2, 2)          # 2 nodes x 2 GPUs
    slots.append(Slot(4, 0, 2, 1))    # node 1 gets a 3rd GPU
    slot_tuples = [(s.rank, s.local_id, s.cuda, s.node) for s in slots]
    pp = py_placement("trivial", size, radius, slots)
    assert pp.uniform_nodes is False
    for rank in range(5):
        cpp = _C.cpp_plan(_C.Vec3(*size), radius, rank, slot_tuples, "trivial")
        assert tuple(cpp["dim"]) == pp.dim()
        plan = plan_exchange(pp, radius, rank)
        assert len(cpp["translates"]) == len(plan.translates)
        assert len(cpp["sends"]) == len(plan.sends)
        assert len(cpp["recvs"]) == len(plan.recvs)
        for (peer, sg, dg, li, msgs, tag), it in zip(cpp["sends"], plan.sends):
            assert (peer, sg, dg, li) == (it.peer_rank, it.src_gid, it.dst_gid, it.local_id)
