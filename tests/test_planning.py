"""Exchange-plan invariants (no GPU, no torch.distributed): the sends one
rank plans must exactly mirror the recvs its peers plan, with identical
wire layouts (this is what makes RCCL's order-based matching correct)."""
from stencil_amd import _C
from stencil_amd.parallel.placement import Slot, TrivialPlacement
from stencil_amd.parallel.planning import plan_exchange, wire_layout


def make_placement(size, radius, n_ranks, gpus_per_rank=1):
    slots = [
        Slot(r, li, li, 0) for r in range(n_ranks) for li in range(gpus_per_rank)
    ]
    return TrivialPlacement(size, radius, slots)


def test_single_rank_all_translates():
    r = _C.Radius.constant(1)
    p = make_placement((32, 32, 32), r, 1, 8)
    plan = plan_exchange(p, r, 0)
    assert not plan.sends and not plan.recvs
    assert len(plan.translates) == 8 * 26  # every dir of every domain


def test_self_wrap_single_domain():
    r = _C.Radius.constant(2)
    p = make_placement((16, 16, 16), r, 1, 1)
    plan = plan_exchange(p, r, 0)
    assert len(plan.translates) == 26
    for t in plan.translates:
        assert t.src_local == 0 and t.dst_local == 0


def test_send_recv_symmetry_two_ranks():
    r = _C.Radius.constant(1)
    p = make_placement((20, 20, 20), r, 2, 1)
    plan0 = plan_exchange(p, r, 0)
    plan1 = plan_exchange(p, r, 1)
    # every send of rank0 to rank1 has a matching recv on rank1
    s0 = {(s.src_gid, s.dst_gid): s for s in plan0.sends if s.peer_rank == 1}
    r1 = {(x.src_gid, x.dst_gid): x for x in plan1.recvs if x.peer_rank == 0}
    assert set(s0) == set(r1) and len(s0) > 0
    es = [4, 8]
    for key in s0:
        send, recv = s0[key], r1[key]
        dirs_s = [m.dir for m in send.messages]
        dirs_r = [m.dir for m in recv.messages]
        assert dirs_s == dirs_r
        assert [m.ext for m in send.messages] == [m.ext for m in recv.messages]
        ts, cs = wire_layout(send.messages, es)
        tr, cr = wire_layout(recv.messages, es)
        assert ts == tr and cs == cr


def test_asymmetric_radius_extents():
    r = _C.Radius.constant(1)
    r.set_dir(1, 0, 0, 2)  # +x reaches 2
    p = make_placement((24, 8, 8), r, 2, 1)  # splits x into 2
    assert p.dim() == (2, 1, 1)
    plan0 = plan_exchange(p, r, 0)
    # a send in +x fills the receiver's -x halo: extent.x == radius(-x) == 1
    px = [s for s in plan0.sends for m in s.messages if m.dir == (1, 0, 0)]
    assert px, "expected a +x send"
    for s in plan0.sends:
        for m in s.messages:
            if m.dir == (1, 0, 0):
                assert m.ext[0] == 1
            if m.dir == (-1, 0, 0):
                assert m.ext[0] == 2


def test_zero_radius_dir_suppresses_messages():
    r = _C.Radius.constant(0)
    r.set_dir(-1, 0, 0, 1)  # only -x halo exists -> only +x sends
    p = make_placement((16, 16, 16), r, 2, 1)
    plan0 = plan_exchange(p, r, 0)
    dirs = {m.dir for s in plan0.sends for m in s.messages}
    dirs |= {t.dir for t in plan0.translates}
    assert dirs == {(1, 0, 0)}


def test_wire_layout_alignment():
    from stencil_amd.parallel.planning import Message

    msgs = [Message((1, 0, 0), 0, 1, (1, 3, 5)), Message((0, 1, 0), 0, 1, (7, 1, 5))]
    total, chunks = wire_layout(msgs, [4, 1, 8])
    offs = [c[2] for c in chunks]
    assert offs == sorted(offs)
    for _, _, off, _ in chunks:
        assert off % 16 == 0
    assert total % 16 == 0
    # chunks must not overlap
    for (a, b) in zip(chunks, chunks[1:]):
        assert a[2] + a[3] <= b[2]


def test_wire_layout_group_subset_deterministic():
    """group exchanges: a qis subset must produce identical layouts on
    both sides regardless of the subset's given order, and cover only the
    subset's quantities (the RCCL wire matches by order, not tags)"""
    from stencil_amd.parallel.planning import Message

    msgs = [Message((1, 0, 0), 0, 1, (2, 3, 5)), Message((0, 0, 1), 0, 1, (7, 6, 1))]
    es = [4, 8, 4, 8]
    t1, c1 = wire_layout(msgs, es, qis=[3, 1])
    t2, c2 = wire_layout(msgs, es, qis=[1, 3])
    assert t1 == t2 and c1 == c2
    qset = {qi for _, qi, _, _ in c1}
    assert qset == {1, 3}
    # full-set layout differs (more bytes) but stays consistent
    tf, cf = wire_layout(msgs, es, qis=None)
    assert tf > t1
    assert {qi for _, qi, _, _ in cf} == {0, 1, 2, 3}
