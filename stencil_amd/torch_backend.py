"""Torch reference backend.

A plain-PyTorch implementation of the same exchange plan, used as
1) the fp32/fp64 numerics reference that every HIP kernel is tested
   against, and 2) the CPU path for multi-process (gloo) tests of the
   distributed planning/transport logic on machines without GPUs.
Not a performance path.
"""
from __future__ import annotations

from typing import Tuple

import torch

from . import _C
from .parallel.planning import ExchangePlan, pair_seq_tags, wire_layout

Vec = Tuple[int, int, int]

_DTYPES = {1: torch.uint8, 2: torch.int16, 4: torch.float32, 8: torch.float64}


def _vec3(t):
    return _C.Vec3(int(t[0]), int(t[1]), int(t[2]))


class _TorchDomain:
    """tensor-backed LocalDomain mirror (geometry via the native statics)"""

    def __init__(self, size: Vec, origin: Vec, device: str, data_defs, radius):
        self.size = tuple(size)
        self.origin = tuple(origin)
        self.device = device
        self.radius = radius
        self.elem_sizes = [es for es, _ in data_defs]
        raw = self.raw_size()
        self.curr = []
        self.next = []
        for es, _name in data_defs:
            if es not in _DTYPES:
                raise ValueError(f"torch backend supports elem sizes {list(_DTYPES)}, got {es}")
            shape = (raw[2], raw[1], raw[0])  # (z, y, x)
            self.curr.append(torch.zeros(shape, dtype=_DTYPES[es], device=device))
            self.next.append(torch.zeros(shape, dtype=_DTYPES[es], device=device))

    def raw_size(self) -> Vec:
        r = self.radius
        return (
            self.size[0] + r.x(-1) + r.x(1),
            self.size[1] + r.y(-1) + r.y(1),
            self.size[2] + r.z(-1) + r.z(1),
        )

    def halo_pos(self, d: Vec, halo: bool) -> Vec:
        return _C.halo_pos(_vec3(d), _vec3(self.size), self.radius, halo).tuple()

    def halo_extent(self, d: Vec) -> Vec:
        return _C.halo_extent(_vec3(d), _vec3(self.size), self.radius).tuple()

    def full_lo(self) -> Vec:
        r = self.radius
        return (self.origin[0] - r.x(-1), self.origin[1] - r.y(-1), self.origin[2] - r.z(-1))

    @staticmethod
    def _sl(pos: Vec, ext: Vec):
        return (
            slice(pos[2], pos[2] + ext[2]),
            slice(pos[1], pos[1] + ext[1]),
            slice(pos[0], pos[0] + ext[0]),
        )

    def region(self, qi: int, pos: Vec, ext: Vec, from_next=False) -> torch.Tensor:
        t = (self.next if from_next else self.curr)[qi]
        return t[self._sl(pos, ext)]


class TorchBackend:
    def __init__(self, domain_specs, data_defs, radius, device: str = "cpu", groups=None):
        self.radius = radius
        self.data_defs = list(data_defs)
        self.groups = groups if groups is not None else [list(range(len(data_defs)))]
        if device.startswith("cuda") and len({c for _, _, c in domain_specs}) > 1:
            # one tensor device per local domain
            self.domains = [
                _TorchDomain(s, o, f"cuda:{c}", data_defs, radius) for s, o, c in domain_specs
            ]
        else:
            self.domains = [
                _TorchDomain(s, o, device, data_defs, radius) for s, o, c in domain_specs
            ]
        ng = len(self.groups)
        self._translates = [[] for _ in range(ng)]
        self._send_bufs = [[] for _ in range(ng)]  # per group
        self._recv_bufs = [[] for _ in range(ng)]

    def register_plan(self, plan: ExchangePlan, ctx=None):
        elem_sizes = [es for es, _ in self.data_defs]
        seq = pair_seq_tags(plan)
        ng = len(self.groups)
        for g, qis in enumerate(self.groups):
            for t in plan.translates:
                src = self.domains[t.src_local]
                dst = self.domains[t.dst_local]
                nd = tuple(-c for c in t.dir)
                self._translates[g].append(
                    (t.src_local, t.dst_local, src.halo_pos(t.dir, False),
                     dst.halo_pos(nd, True), t.ext, sorted(qis))
                )
            for item, is_send in [(s, True) for s in plan.sends] + [(r, False) for r in plan.recvs]:
                total, chunks = wire_layout(item.messages, elem_sizes, qis)
                dom = self.domains[item.local_id]
                buf = torch.zeros(total, dtype=torch.uint8, device=dom.device)
                entries = []
                for mi, qi, off, nbytes in chunks:
                    m = item.messages[mi]
                    nd = tuple(-c for c in m.dir)
                    pos = dom.halo_pos(m.dir, False) if is_send else dom.halo_pos(nd, True)
                    entries.append((off, nbytes, item.local_id, pos, m.ext, qi))
                tag = seq[(item.peer_rank, item.src_gid, item.dst_gid)] * ng + g
                rec = (buf, item.peer_rank, tag, entries)
                (self._send_bufs[g] if is_send else self._recv_bufs[g]).append(rec)

    def _pack(self, g):
        for buf, _peer, _tag, entries in self._send_bufs[g]:
            for off, nbytes, li, pos, ext, qi in entries:
                reg = self.domains[li].region(qi, pos, ext).contiguous()
                buf[off : off + nbytes] = reg.view(-1).view(torch.uint8)

    def _unpack(self, g):
        for buf, _peer, _tag, entries in self._recv_bufs[g]:
            for off, nbytes, li, pos, ext, qi in entries:
                dom = self.domains[li]
                dtype = dom.curr[qi].dtype
                reg = buf[off : off + nbytes].view(dtype).reshape(ext[2], ext[1], ext[0])
                dom.region(qi, pos, ext).copy_(reg)

    def exchange(self, group: int = 0):
        g = group
        for sl, dl, spos, dpos, ext, qis in self._translates[g]:
            src = self.domains[sl]
            dst = self.domains[dl]
            for qi in qis:
                dst.region(qi, dpos, ext).copy_(src.region(qi, spos, ext))
        if self._send_bufs[g] or self._recv_bufs[g]:
            import torch.distributed as dist

            self._pack(g)
            ops = [
                dist.P2POp(dist.isend, buf, peer, tag=tag)
                for buf, peer, tag, _ in self._send_bufs[g]
            ]
            ops += [
                dist.P2POp(dist.irecv, buf, peer, tag=tag)
                for buf, peer, tag, _ in self._recv_bufs[g]
            ]
            for w in dist.batch_isend_irecv(ops):
                w.wait()
            self._unpack(g)

    def swap(self):
        for d in self.domains:
            d.curr, d.next = d.next, d.curr

    def sync(self):
        pass

    # ---- helpers mirroring NativeBackend ----
    def read_region(self, li, pos, ext, qi, from_next=False) -> bytes:
        reg = self.domains[li].region(qi, pos, ext, from_next).contiguous().cpu()
        return reg.view(-1).view(torch.uint8).numpy().tobytes()

    def write_region(self, li, data: bytes, pos, ext, qi, to_next=False):
        dom = self.domains[li]
        dtype = dom.curr[qi].dtype
        t = torch.frombuffer(bytearray(data), dtype=torch.uint8).view(dtype)
        t = t.reshape(ext[2], ext[1], ext[0]).to(dom.device)
        dom.region(qi, pos, ext, to_next).copy_(t)

    def fill_f32(self, li, qi, region_lo, region_hi, value, next_buf):
        dom = self.domains[li]
        flo = dom.full_lo()
        pos = tuple(region_lo[i] - flo[i] for i in range(3))
        ext = tuple(region_hi[i] - region_lo[i] for i in range(3))
        dom.region(qi, pos, ext, next_buf).fill_(value)

    def jacobi_step(self, li, qi, region_lo, region_hi, c_lo, c_hi, stream_id=0,
                    extend_vec=0):  # extend_vec: native-only fast-path hint, ignored here
        """reference 7-point Jacobi with hot/cold spheres (fp32)"""
        dom = self.domains[li]
        flo = dom.full_lo()
        pos = tuple(region_lo[i] - flo[i] for i in range(3))
        ext = tuple(region_hi[i] - region_lo[i] for i in range(3))
        src = dom.curr[qi]
        sl = dom._sl(pos, ext)

        def sh(dx, dy, dz):
            p = (pos[0] + dx, pos[1] + dy, pos[2] + dz)
            return src[dom._sl(p, ext)]

        avg = (sh(1, 0, 0) + sh(-1, 0, 0) + sh(0, 1, 0) + sh(0, -1, 0) + sh(0, 0, 1) + sh(0, 0, -1)) / 6.0

        # hot/cold spheres (truncated-int sqrt distance like the reference)
        device = src.device
        zz = torch.arange(region_lo[2], region_hi[2], device=device).view(-1, 1, 1)
        yy = torch.arange(region_lo[1], region_hi[1], device=device).view(1, -1, 1)
        xx = torch.arange(region_lo[0], region_hi[0], device=device).view(1, 1, -1)
        cw = c_hi[0] - c_lo[0]
        hot = (c_lo[0] + cw // 3, (c_lo[1] + c_hi[1]) // 2, (c_lo[2] + c_hi[2]) // 2)
        cold = (c_lo[0] + cw * 2 // 3, hot[1], hot[2])
        r = cw // 10

        def mask(center):
            d2 = (xx - center[0]) ** 2 + (yy - center[1]) ** 2 + (zz - center[2]) ** 2
            return torch.sqrt(d2.float()).long() <= r

        out = torch.where(mask(hot), torch.ones_like(avg), avg)
        out = torch.where(mask(cold), torch.zeros_like(avg), out)
        dom.next[qi][sl] = out

    def sync_compute(self):
        pass
