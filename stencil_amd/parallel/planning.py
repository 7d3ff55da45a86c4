"""Halo-exchange message planning (backend-agnostic, pure logic).

Mirrors the reference's plan loop (reference: src/stencil.cu:327-464) and
wire conventions:
- a send in direction d happens iff radius(-d) != 0,
- the send extent is the RECEIVER's -d halo extent (halo_extent(-d) with the
  receiver's size; reference src/packer.cu:78-82),
- the source region is halo_pos(d, interior) of the sender, the destination
  region is halo_pos(-d, halo) of the receiver.

Cross-rank messages between one (src subdomain, dst subdomain) pair are
batched into ONE contiguous buffer covering every direction and every
quantity (the reference's per-direction quantity batching, generalized), so
RCCL sees few, large point-to-point transfers over xGMI.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Tuple

from .. import _C
from .topology import dir_key

Vec = Tuple[int, int, int]


def _vec3(t) -> "_C.Vec3":
    return _C.Vec3(int(t[0]), int(t[1]), int(t[2]))


@dataclass(frozen=True)
class Message:
    """one directional halo message between two subdomains"""

    dir: Vec
    src_gid: int
    dst_gid: int
    ext: Vec  # element extent (equals receiver's halo_extent(-dir))

    def volume(self) -> int:
        return self.ext[0] * self.ext[1] * self.ext[2]


@dataclass
class TranslatePlanItem:
    """same-rank direct-write copy (handled by translate kernels)"""

    src_local: int
    dst_local: int
    dir: Vec
    ext: Vec


@dataclass
class WirePlanItem:
    """cross-rank packed transfer for one (src subdomain, dst subdomain) pair"""

    peer_rank: int
    src_gid: int
    dst_gid: int
    local_id: int  # my local domain involved (src for sends, dst for recvs)
    messages: List[Message] = field(default_factory=list)


@dataclass
class ExchangePlan:
    translates: List[TranslatePlanItem] = field(default_factory=list)
    sends: List[WirePlanItem] = field(default_factory=list)
    recvs: List[WirePlanItem] = field(default_factory=list)


def plan_exchange(placement, radius: "_C.Radius", rank: int) -> ExchangePlan:
    """plan all messages this rank participates in"""
    plan = ExchangePlan()
    dim = placement.dim()
    sends: Dict[Tuple[int, int, int], WirePlanItem] = {}
    recvs: Dict[Tuple[int, int, int], WirePlanItem] = {}

    n_local = placement.num_local(rank)
    for li in range(n_local):
        my_idx = placement.get_idx(rank, li)
        my_gid = placement.linearize(my_idx)
        for dz in (-1, 0, 1):
            for dy in (-1, 0, 1):
                for dx in (-1, 0, 1):
                    d = (dx, dy, dz)
                    if d == (0, 0, 0):
                        continue
                    if radius.dir(-dx, -dy, -dz) == 0:
                        continue
                    neg = (-dx, -dy, -dz)

                    # --- send to the neighbor at +d ---
                    dst_idx = tuple((my_idx[i] + d[i]) % dim[i] for i in range(3))
                    dst_gid = placement.linearize(dst_idx)
                    dst_rank = placement.get_rank(dst_idx)
                    dst_size = placement.subdomain_size(dst_idx)
                    s_ext = _C.halo_extent(_vec3(neg), _vec3(dst_size), radius).tuple()
                    if s_ext[0] * s_ext[1] * s_ext[2] == 0:
                        # degenerate: a diagonal radius is set but a face
                        # radius of one of its components is 0, so the halo
                        # region has no volume (the reference fatals here,
                        # src/packer.cu:83; we skip the empty message)
                        continue
                    msg = Message(d, my_gid, dst_gid, s_ext)
                    if dst_rank == rank:
                        plan.translates.append(
                            TranslatePlanItem(li, placement.get_subdomain_id(dst_idx), d, s_ext)
                        )
                    else:
                        key = (dst_rank, my_gid, dst_gid)
                        if key not in sends:
                            sends[key] = WirePlanItem(dst_rank, my_gid, dst_gid, li)
                        sends[key].messages.append(msg)

                    # --- recv from the neighbor at -d (message travels in +d) ---
                    src_idx = tuple((my_idx[i] + neg[i]) % dim[i] for i in range(3))
                    src_gid = placement.linearize(src_idx)
                    src_rank = placement.get_rank(src_idx)
                    my_size = placement.subdomain_size(my_idx)
                    r_ext = _C.halo_extent(_vec3(neg), _vec3(my_size), radius).tuple()
                    if src_rank != rank and r_ext[0] * r_ext[1] * r_ext[2] > 0:
                        # (same-rank recvs are covered by the send loop;
                        # zero-volume messages skipped as above)
                        key = (src_rank, src_gid, my_gid)
                        if key not in recvs:
                            recvs[key] = WirePlanItem(src_rank, src_gid, my_gid, li)
                        recvs[key].messages.append(Message(d, src_gid, my_gid, r_ext))

    def _sorted(items: Dict) -> List[WirePlanItem]:
        out = []
        for key in sorted(items.keys()):
            item = items[key]
            item.messages.sort(key=lambda m: dir_key(m.dir))
            out.append(item)
        return out

    plan.sends = _sorted(sends)
    plan.recvs = _sorted(recvs)
    return plan


def wire_layout(messages: List[Message], elem_sizes: List[int], qis=None
                ) -> Tuple[int, List[Tuple[int, int, int, int]]]:
    """byte layout of one packed buffer: per message (sorted by direction),
    per quantity of the exchange group (sorted indices; None = all), a
    16 B-aligned chunk. Returns (total_bytes, chunks) with chunks =
    [(msg_index, qi, offset, nbytes)]. Both ranks compute this
    identically, so it is the wire format."""
    if qis is None:
        qis = range(len(elem_sizes))
    chunks = []
    off = 0
    for mi, m in enumerate(messages):
        for qi in sorted(qis):
            off = (off + 15) // 16 * 16
            nbytes = elem_sizes[qi] * m.volume()
            chunks.append((mi, qi, off, nbytes))
            off += nbytes
    return (off + 15) // 16 * 16, chunks


def wire_layout_pairs(pairs, elem_sizes: List[int]
                      ) -> Tuple[int, List[Tuple["Message", int, int, int]]]:
    """byte layout of one packed buffer over explicit (message, quantity)
    pairs (the staged thin-chunk IPC path, where thin-ness is decided per
    quantity). Both ranks enumerate the pairs in the same deterministic
    order, so the layout is the wire format. Returns (total_bytes, chunks)
    with chunks = [(message, qi, offset, nbytes)]."""
    chunks = []
    off = 0
    for m, qi in pairs:
        off = (off + 15) // 16 * 16
        nbytes = elem_sizes[qi] * m.volume()
        chunks.append((m, qi, off, nbytes))
        off += nbytes
    return (off + 15) // 16 * 16, chunks


def pair_seq_tags(plan: ExchangePlan) -> Dict[Tuple[int, int, int], int]:
    """injective per-rank-pair tag for every wire transfer in the plan:
    the index of (src_gid, dst_gid) in the sorted set of all transfers
    (either direction) between this rank and that peer. Sender and
    receiver enumerate the same key set for a given rank pair, so the
    indices agree on both sides, and they are collision-free by
    construction for any partition size (the round-1 arithmetic tag
    `(src_gid*4096+dst_gid) % 2^20` aliased past 4096 gids; RCCL ignores
    tags but gloo matches on them)."""
    by_peer: Dict[int, set] = {}
    for item in list(plan.sends) + list(plan.recvs):
        by_peer.setdefault(item.peer_rank, set()).add((item.src_gid, item.dst_gid))
    seq: Dict[Tuple[int, int, int], int] = {}
    for peer, keys in by_peer.items():
        for i, (sg, dg) in enumerate(sorted(keys)):
            seq[(peer, sg, dg)] = i
    return seq
