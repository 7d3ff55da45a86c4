// pybind11 bindings for the stencil_amd native core.
//
// The native library is self-contained C++/HIP (no libtorch dependency);
// staging buffers are exported to Python through DLPack capsules so
// torch.from_dlpack can alias them for RCCL (torch.distributed) transport.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstring>
#include <memory>

#include "stencil_amd/core.hpp"
#include "stencil_amd/distributed.hpp"
#include "stencil_amd/domain.hpp"
#include "stencil_amd/engine.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/ops.hpp"
#include "stencil_amd/partition.hpp"
#include "stencil_amd/placement.hpp"
#include "stencil_amd/planning.hpp"
#include "stencil_amd/qap.hpp"
#include "stencil_amd/topo.hpp"
#include "stencil_amd/wire.hpp"

namespace py = pybind11;
using namespace stencil_amd;

namespace {

//// minimal DLPack (v0.8 ABI) ////
typedef struct {
  int32_t device_type;
  int32_t device_id;
} DLDevice_;
typedef struct {
  uint8_t code;
  uint8_t bits;
  uint16_t lanes;
} DLDataType_;
typedef struct {
  void *data;
  DLDevice_ device;
  int32_t ndim;
  DLDataType_ dtype;
  int64_t *shape;
  int64_t *strides;
  uint64_t byte_offset;
} DLTensor_;
struct DLManagedTensor_ {
  DLTensor_ dl_tensor;
  void *manager_ctx;
  void (*deleter)(DLManagedTensor_ *);
};
constexpr int32_t kDLROCM = 10;

struct DLHolder {
  DLManagedTensor_ mt;
  int64_t shape[1];
};

// 1-D uint8 view of `bytes` device bytes at `ptr` on HIP device `dev`.
// The engine owns the memory; the capsule only owns the descriptor.
py::capsule make_dlpack_u8(uintptr_t ptr, int64_t bytes, int dev) {
  auto *h = new DLHolder();
  h->shape[0] = bytes;
  h->mt.dl_tensor.data = (void *)ptr;
  h->mt.dl_tensor.device = {kDLROCM, dev};
  h->mt.dl_tensor.ndim = 1;
  h->mt.dl_tensor.dtype = {1 /*kDLUInt*/, 8, 1};
  h->mt.dl_tensor.shape = h->shape;
  h->mt.dl_tensor.strides = nullptr;
  h->mt.dl_tensor.byte_offset = 0;
  h->mt.manager_ctx = h;
  h->mt.deleter = [](DLManagedTensor_ *mt) { delete (DLHolder *)mt->manager_ctx; };
  return py::capsule(&h->mt, "dltensor", [](PyObject *cap) {
    if (PyCapsule_IsValid(cap, "dltensor")) {
      auto *mt = (DLManagedTensor_ *)PyCapsule_GetPointer(cap, "dltensor");
      if (mt && mt->deleter) mt->deleter(mt);
    }
  });
}

} // namespace

PYBIND11_MODULE(_C, m) {
  m.doc() = "stencil_amd native core (MI355X / gfx950)";

  py::class_<Vec3>(m, "Vec3")
      .def(py::init<int64_t, int64_t, int64_t>())
      .def_readwrite("x", &Vec3::x)
      .def_readwrite("y", &Vec3::y)
      .def_readwrite("z", &Vec3::z)
      .def("__eq__", [](const Vec3 &a, const Vec3 &b) { return a == b; })
      .def("__lt__", [](const Vec3 &a, const Vec3 &b) { return a < b; })
      .def("__add__", [](const Vec3 &a, const Vec3 &b) { return a + b; })
      .def("__sub__", [](const Vec3 &a, const Vec3 &b) { return a - b; })
      .def("__mul__", [](const Vec3 &a, const Vec3 &b) { return a * b; })
      .def("__mod__", [](const Vec3 &a, const Vec3 &b) { return a % b; })
      .def("__neg__", [](const Vec3 &a) { return -a; })
      .def("__hash__", [](const Vec3 &a) { return py::hash(py::make_tuple(a.x, a.y, a.z)); })
      .def("wrap", &Vec3::wrap)
      .def("flatten", &Vec3::flatten)
      .def("tuple", [](const Vec3 &a) { return py::make_tuple(a.x, a.y, a.z); })
      .def("__repr__", &Vec3::str);

  py::class_<Rect3>(m, "Rect3")
      .def(py::init<const Vec3 &, const Vec3 &>())
      .def_readwrite("lo", &Rect3::lo)
      .def_readwrite("hi", &Rect3::hi)
      .def("extent", &Rect3::extent)
      .def("volume", &Rect3::volume)
      .def("contains", &Rect3::contains)
      .def("__eq__", [](const Rect3 &a, const Rect3 &b) { return a == b; })
      .def("__repr__", &Rect3::str);

  py::class_<Radius>(m, "Radius")
      .def(py::init<>())
      .def_static("constant", &Radius::constant)
      .def_static("face_edge_corner", &Radius::face_edge_corner)
      .def("dir", [](const Radius &r, int x, int y, int z) { return r.dir(x, y, z); })
      .def("set_dir", [](Radius &r, int x, int y, int z, int64_t v) { r.dir(x, y, z) = v; })
      .def("x", &Radius::x)
      .def("y", &Radius::y)
      .def("z", &Radius::z)
      .def("__eq__", [](const Radius &a, const Radius &b) { return a == b; });

  m.def("prime_factors", &prime_factors);
  m.def("div_ceil", &div_ceil);

  py::class_<RankPartition>(m, "RankPartition")
      .def(py::init<const Vec3 &, int64_t>())
      .def("dim", &RankPartition::dim)
      .def("subdomain_size", &RankPartition::subdomain_size)
      .def("subdomain_origin", &RankPartition::subdomain_origin)
      .def("linearize", &RankPartition::linearize)
      .def("dimensionize", &RankPartition::dimensionize);

  py::class_<NodePartition>(m, "NodePartition")
      .def(py::init<const Vec3 &, const Radius &, int64_t, int64_t>())
      .def("sys_dim", &NodePartition::sys_dim)
      .def("node_dim", &NodePartition::node_dim)
      .def("dim", &NodePartition::dim)
      .def("subdomain_size", &NodePartition::subdomain_size)
      .def("subdomain_origin", &NodePartition::subdomain_origin);

  py::class_<SqMat>(m, "SqMat")
      .def(py::init<int64_t, double>(), py::arg("n"), py::arg("fill") = 0.0)
      .def("set", [](SqMat &mm, int64_t i, int64_t j, double v) { mm.at(i, j) = v; })
      .def("get", [](const SqMat &mm, int64_t i, int64_t j) { return mm.at(i, j); });
  m.def("qap_solve", &qap::solve, py::arg("w"), py::arg("d"), py::arg("timeout_sec") = 10.0);
  m.def("qap_cost", &qap::cost);

  py::class_<LocalDomain, std::shared_ptr<LocalDomain>>(m, "LocalDomain")
      .def(py::init<const Vec3 &, const Vec3 &, int>())
      .def("add_data", &LocalDomain::add_data, py::arg("elem_size"), py::arg("name") = "")
      .def("set_radius", &LocalDomain::set_radius)
      .def("realize", &LocalDomain::realize)
      .def("swap", &LocalDomain::swap)
      .def("halo_pos", (Vec3(LocalDomain::*)(const Vec3 &, bool) const) & LocalDomain::halo_pos)
      .def("halo_extent", (Vec3(LocalDomain::*)(const Vec3 &) const) & LocalDomain::halo_extent)
      .def("halo_coords", &LocalDomain::halo_coords)
      .def("halo_bytes", &LocalDomain::halo_bytes)
      .def("compute_region", &LocalDomain::compute_region)
      .def("full_region", &LocalDomain::full_region)
      .def("raw_size", &LocalDomain::raw_size)
      .def("size", &LocalDomain::size)
      .def("origin", &LocalDomain::origin)
      .def("gpu", &LocalDomain::gpu)
      .def("num_data", &LocalDomain::num_data)
      .def("elem_size", &LocalDomain::elem_size)
      .def("region_to_host",
           [](const LocalDomain &d, const Vec3 &pos, const Vec3 &ext, int64_t qi, bool fromNext) {
             // uninitialized PyBytes + direct fill: the round-2 pinned
             // bounce was bottlenecked by the std::string zero-fill and
             // the py::bytes copy (two extra full host passes)
             const int64_t n = ext.flatten() * d.elem_size(qi);
             py::bytes out = py::reinterpret_steal<py::bytes>(
                 PyBytes_FromStringAndSize(nullptr, (Py_ssize_t)n));
             if (!out) throw std::bad_alloc();
             d.region_to_host(PyBytes_AS_STRING(out.ptr()), pos, ext, qi, fromNext);
             return out;
           },
           py::arg("pos"), py::arg("ext"), py::arg("qi"), py::arg("from_next") = false)
      .def("region_to_host_into",
           [](const LocalDomain &d, py::buffer out, const Vec3 &pos, const Vec3 &ext, int64_t qi,
              bool fromNext) {
             const py::buffer_info info = out.request(true);
             if (info.size * info.itemsize != ext.flatten() * d.elem_size(qi))
               throw std::runtime_error("region_to_host_into: size mismatch");
             d.region_to_host(info.ptr, pos, ext, qi, fromNext);
           },
           py::arg("out"), py::arg("pos"), py::arg("ext"), py::arg("qi"),
           py::arg("from_next") = false)
      .def("curr_pitch", [](const LocalDomain &d, int64_t qi) { return d.curr(qi).pitch; })
      .def("pad_bytes", &LocalDomain::pad_bytes)
      .def("curr_ysize", [](const LocalDomain &d, int64_t qi) { return d.curr(qi).ysize; })
      .def("ipc_handle",
           [](const LocalDomain &d, int64_t qi, bool next) { return py::bytes(d.ipc_handle(qi, next)); },
           py::arg("qi"), py::arg("next") = false)
      .def("region_from_host",
           [](const LocalDomain &d, py::buffer data, const Vec3 &pos, const Vec3 &ext, int64_t qi,
              bool toNext) {
             // buffer protocol (bytes or a contiguous array), no copy
             const py::buffer_info info = data.request();
             if (info.size * info.itemsize != ext.flatten() * d.elem_size(qi))
               throw std::runtime_error("region_from_host: size mismatch");
             d.region_from_host(info.ptr, pos, ext, qi, toNext);
           },
           py::arg("data"), py::arg("pos"), py::arg("ext"), py::arg("qi"),
           py::arg("to_next") = false);

  // module-level geometry statics (CPU-testable without a GPU)
  m.def("halo_pos", [](const Vec3 &dir, const Vec3 &sz, const Radius &r, bool halo) {
    return LocalDomain::halo_pos(dir, sz, r, halo);
  });
  m.def("halo_extent", [](const Vec3 &dir, const Vec3 &sz, const Radius &r) {
    return LocalDomain::halo_extent(dir, sz, r);
  });

  py::class_<ExchangeEngine>(m, "ExchangeEngine")
      .def(py::init<std::vector<std::shared_ptr<LocalDomain>>>())
      .def("enable_peer_all", &ExchangeEngine::enable_peer_all)
      .def_static("can_access_peer", &ExchangeEngine::can_access_peer)
      .def("add_translate", &ExchangeEngine::add_translate, py::arg("src"), py::arg("dst"),
           py::arg("src_pos"), py::arg("dst_pos"), py::arg("ext"), py::arg("group") = 0,
           py::arg("qis") = std::vector<int64_t>{})
      .def("create_remote_view",
           [](ExchangeEngine &e, int openDev, const std::vector<py::bytes> &cur,
              const std::vector<py::bytes> &nxt, const std::vector<int64_t> &pitches,
              const std::vector<int64_t> &ysizes, const std::vector<int64_t> &es,
              const std::vector<int64_t> &pads) {
             std::vector<std::string> c, n;
             for (auto &b : cur) c.push_back(b);
             for (auto &b : nxt) n.push_back(b);
             return e.create_remote_view(openDev, c, n, pitches, ysizes, es, pads);
           })
      .def("add_translate_view", &ExchangeEngine::add_translate_view, py::arg("src"),
           py::arg("view"), py::arg("src_pos"), py::arg("dst_pos"), py::arg("ext"),
           py::arg("group") = 0, py::arg("qis") = std::vector<int64_t>{})
      .def("flip_views", &ExchangeEngine::flip_views)
      .def("create_buffer", &ExchangeEngine::create_buffer)
      .def("add_pack", &ExchangeEngine::add_pack, py::arg("dom"), py::arg("buf"),
           py::arg("offset"), py::arg("pos"), py::arg("ext"), py::arg("qi"), py::arg("group") = 0)
      .def("add_unpack", &ExchangeEngine::add_unpack, py::arg("dom"), py::arg("buf"),
           py::arg("offset"), py::arg("pos"), py::arg("ext"), py::arg("qi"), py::arg("group") = 0)
      .def("buffer_ipc_handle",
           [](ExchangeEngine &e, int64_t buf) { return py::bytes(e.buffer_ipc_handle(buf)); })
      .def("open_remote_buffer",
           [](ExchangeEngine &e, int openDev, py::bytes handle, int64_t bytes) {
             return e.open_remote_buffer(openDev, std::string(handle), bytes);
           })
      .def("finalize", &ExchangeEngine::finalize)
      .def("launch_translates", &ExchangeEngine::launch_translates, py::arg("group") = 0)
      .def("launch_packs", &ExchangeEngine::launch_packs, py::arg("group") = 0)
      .def("launch_unpacks", &ExchangeEngine::launch_unpacks, py::arg("group") = 0)
      .def("fence_packs_unpacks", &ExchangeEngine::fence_packs_unpacks, py::arg("group") = 0)
      .def("sync_translates", &ExchangeEngine::sync_translates)
      .def("sync_packs", &ExchangeEngine::sync_packs)
      .def("sync_all", &ExchangeEngine::sync_all)
      .def("sync_compute", &ExchangeEngine::sync_compute)
      .def("compute_stream_handle", &ExchangeEngine::compute_stream_handle)
      .def("pack_stream_handle", &ExchangeEngine::pack_stream_handle)
      .def("buffer_ptr", &ExchangeEngine::buffer_ptr)
      .def("buffer_bytes", &ExchangeEngine::buffer_bytes)
      .def("buffer_device", &ExchangeEngine::buffer_device)
      .def("num_domains", &ExchangeEngine::num_domains)
      .def("buffer_dlpack", [](ExchangeEngine &e, int64_t buf) {
        return make_dlpack_u8(e.buffer_ptr(buf), e.buffer_bytes(buf), e.buffer_device(buf));
      })
      .def("buffer_to_host",
           [](ExchangeEngine &e, int64_t buf) {
             std::string out(e.buffer_bytes(buf), '\0');
             STENCIL_HIP(hipSetDevice(e.buffer_device(buf)));
             STENCIL_HIP(hipMemcpy(out.data(), (void *)e.buffer_ptr(buf), out.size(),
                                   hipMemcpyDeviceToHost));
             return py::bytes(out);
           })
      .def("buffer_from_host", [](ExchangeEngine &e, int64_t buf, py::bytes data) {
        std::string s = data;
        if ((int64_t)s.size() != e.buffer_bytes(buf))
          throw std::runtime_error("buffer_from_host: size mismatch");
        STENCIL_HIP(hipSetDevice(e.buffer_device(buf)));
        STENCIL_HIP(
            hipMemcpy((void *)e.buffer_ptr(buf), s.data(), s.size(), hipMemcpyHostToDevice));
      });

  m.def("jacobi_step", &jacobi_step, py::arg("eng"), py::arg("dom"), py::arg("qi"),
        py::arg("region"), py::arg("compute_region"), py::arg("stream_id") = 0,
        py::arg("extend_vec") = 0);
  m.def("fill_f32", &fill_f32);
  m.def("jacobi_graph_create", &jacobi_graph_create, py::arg("eng"), py::arg("dom"),
        py::arg("qi"), py::arg("region"), py::arg("compute_region"));
  m.def("jacobi_graph_create_overlap", &jacobi_graph_create_overlap, py::arg("eng"),
        py::arg("dom"), py::arg("qi"), py::arg("interior"), py::arg("compute_region"),
        py::arg("exteriors"));
  m.def("jacobi_graph_launch", &jacobi_graph_launch, py::arg("handle"), py::arg("n_steps") = 1);
  m.def("jacobi_graph_sync", &jacobi_graph_sync);
  m.def("jacobi_mr_graph_create", &jacobi_mr_graph_create, py::arg("eng"), py::arg("dom"),
        py::arg("qi"), py::arg("interior"), py::arg("compute_region"), py::arg("exteriors"),
        py::arg("extend_vec") = 2);
  m.def("jacobi_mr_graph_stream", &jacobi_mr_graph_stream);
  m.def("jacobi_mr_graph_pre", &jacobi_mr_graph_pre);
  m.def("jacobi_mr_graph_post", &jacobi_mr_graph_post);
  m.def("jacobi_mr_graph_sync", &jacobi_mr_graph_sync);
  m.def("mhd_graph_create", &mhd_graph_create, py::arg("eng"), py::arg("dom"), py::arg("region"),
        py::arg("dt"), py::arg("cf"));
  m.def("mhd_graph_iter", &mhd_graph_iter, py::arg("handle"), py::arg("n_iters") = 1);
  m.def("mhd_graph_sync", &mhd_graph_sync);
  m.def("mhd_mr_graph_create", &mhd_mr_graph_create, py::arg("eng"), py::arg("dom"),
        py::arg("interior"), py::arg("exteriors"), py::arg("dt"), py::arg("cf"));
  m.def("mhd_mr_graph_stream", &mhd_mr_graph_stream);
  m.def("mhd_mr_phase1", &mhd_mr_phase1);
  m.def("mhd_mr_phase2", &mhd_mr_phase2);
  m.def("mhd_mr_phase3", &mhd_mr_phase3);
  m.def("mhd_mr_graph_sync", &mhd_mr_graph_sync);

  py::class_<MhdCoeffs>(m, "MhdCoeffs")
      .def(py::init<>())
      .def_readwrite("dsx", &MhdCoeffs::dsx)
      .def_readwrite("dsy", &MhdCoeffs::dsy)
      .def_readwrite("dsz", &MhdCoeffs::dsz)
      .def_readwrite("cs2", &MhdCoeffs::cs2)
      .def_readwrite("cp_inv", &MhdCoeffs::cp_inv)
      .def_readwrite("nu", &MhdCoeffs::nu)
      .def_readwrite("eta", &MhdCoeffs::eta)
      .def_readwrite("chi", &MhdCoeffs::chi);
  m.def("mhd_substep", &mhd_substep, py::arg("eng"), py::arg("dom"), py::arg("region"),
        py::arg("step"), py::arg("dt"), py::arg("cf"), py::arg("stream_id") = 0);
  m.def("mhd_div_pass", &mhd_div_pass, py::arg("eng"), py::arg("dom"), py::arg("region"),
        py::arg("cf"), py::arg("stream_id") = 0);
  m.def("init_harmonic_f64", &init_harmonic_f64);
  m.def("init_radial_f64", &init_radial_f64);
  py::class_<FieldStats>(m, "FieldStats")
      .def_readonly("min", &FieldStats::min)
      .def_readonly("max", &FieldStats::max)
      .def_readonly("rms", &FieldStats::rms);
  m.def("field_stats", &field_stats, py::arg("eng"), py::arg("dom"), py::arg("qi"),
        py::arg("region"), py::arg("next_buf") = false);

  // C++ orchestrator introspection: run the native placement + planner
  // (placement.hpp/planning.hpp — the ones the C++ DistributedDomain
  // uses) and return plain structures, so tests can pin C++/Python plan
  // parity without a GPU.
  m.def(
      "cpp_plan",
      [](const Vec3 &size, const Radius &radius, int rank,
         const std::vector<std::tuple<int, int, int, int>> &slotTuples,
         const std::string &strategy) {
        std::vector<Slot> slots;
        for (auto &t : slotTuples)
          slots.push_back({std::get<0>(t), std::get<1>(t), std::get<2>(t), std::get<3>(t)});
        auto haloExtent = [](const Vec3 &d, const Vec3 &sz, const Radius &r) {
          return LocalDomain::halo_extent(d, sz, r);
        };
        std::unique_ptr<Placement> p;
        if (strategy == "trivial")
          p = std::make_unique<TrivialPlacement>(size, radius, slots);
        else if (strategy == "node_aware")
          p = std::make_unique<NodeAwarePlacement>(size, radius, slots, haloExtent);
        else
          throw std::runtime_error("cpp_plan: unknown strategy " + strategy);
        const ExchangePlan plan = plan_exchange(*p, radius, rank, haloExtent);
        const auto seq = pair_seq_tags(plan);

        auto tup3 = [](const Vec3 &v) { return py::make_tuple(v.x, v.y, v.z); };
        py::list assign;
        const int64_t n = p->dim().flatten();
        for (int64_t gid = 0; gid < n; ++gid) {
          const Vec3 idx = p->dimensionize(gid);
          assign.append(py::make_tuple(p->get_rank(idx), p->get_subdomain_id(idx),
                                       p->get_cuda(idx)));
        }
        py::list translates;
        for (const auto &t : plan.translates)
          translates.append(py::make_tuple(t.srcLocal, t.dstLocal, tup3(t.dir), tup3(t.ext)));
        auto emit = [&](const std::vector<WirePlanItem> &items) {
          py::list out;
          for (const auto &it : items) {
            py::list msgs;
            for (const auto &m : it.messages)
              msgs.append(py::make_tuple(tup3(m.dir), m.srcGid, m.dstGid, tup3(m.ext)));
            out.append(py::make_tuple(it.peerRank, it.srcGid, it.dstGid, it.localId, msgs,
                                      seq.at({it.peerRank, it.srcGid, it.dstGid})));
          }
          return out;
        };
        py::dict out;
        out["dim"] = tup3(p->dim());
        out["assign"] = assign;
        out["translates"] = translates;
        out["sends"] = emit(plan.sends);
        out["recvs"] = emit(plan.recvs);
        return out;
      },
      py::arg("size"), py::arg("radius"), py::arg("rank"), py::arg("slots"),
      py::arg("strategy") = "node_aware");

  m.def("cpp_wire_layout", [](const std::vector<std::tuple<py::tuple, int64_t, int64_t, py::tuple>> &msgs,
                              const std::vector<int64_t> &elemSizes,
                              const std::vector<int64_t> &qis) {
    std::vector<PlanMessage> ms;
    for (auto &t : msgs) {
      py::tuple d = std::get<0>(t), e = std::get<3>(t);
      ms.push_back({Vec3(d[0].cast<int64_t>(), d[1].cast<int64_t>(), d[2].cast<int64_t>()),
                    std::get<1>(t), std::get<2>(t),
                    Vec3(e[0].cast<int64_t>(), e[1].cast<int64_t>(), e[2].cast<int64_t>())});
    }
    std::vector<WireChunk> chunks;
    const int64_t total = wire_layout(ms, elemSizes, qis, chunks);
    py::list out;
    for (const auto &c : chunks) out.append(py::make_tuple(c.msgIndex, c.qi, c.offset, c.nbytes));
    return py::make_tuple(total, out);
  });

  // The C++ orchestrator itself (distributed.hpp), bound so the GPU test
  // tier can drive it directly (examples/*.cpp are the no-Python proof;
  // this binding pins its behavior under pytest)
  py::class_<DistributedDomain>(m, "CppDistributedDomain")
      .def(py::init<int64_t, int64_t, int64_t>())
      .def("set_radius", (void(DistributedDomain::*)(const Radius &)) & DistributedDomain::set_radius)
      .def("add_data", (int64_t(DistributedDomain::*)(int64_t, const std::string &)) &
                           DistributedDomain::add_data,
           py::arg("elem_size"), py::arg("name") = "")
      .def("set_gpus", &DistributedDomain::set_gpus)
      .def("set_placement_trivial",
           [](DistributedDomain &d) { d.set_placement(PlacementStrategy::Trivial); })
      .def("set_exchange_groups", &DistributedDomain::set_exchange_groups)
      .def("realize", &DistributedDomain::realize)
      .def("exchange", &DistributedDomain::exchange, py::arg("group") = 0)
      .def("swap", &DistributedDomain::swap)
      .def("num_local", &DistributedDomain::num_local)
      .def("local_rect", &DistributedDomain::local_rect)
      .def("get_interior", &DistributedDomain::get_interior)
      .def("get_exterior", &DistributedDomain::get_exterior)
      .def("bytes_translate", &DistributedDomain::bytes_translate)
      .def("bytes_wire", &DistributedDomain::bytes_wire)
      .def("write_paraview", &DistributedDomain::write_paraview)
      .def("setup_times", &DistributedDomain::setup_times)
      .def("domain", &DistributedDomain::domain, py::return_value_policy::reference_internal)
      .def("rank", &DistributedDomain::rank)
      .def("world", &DistributedDomain::world);

  // FileBootstrap (distributed.hpp): the C++ multi-process control plane,
  // bound for CPU-side tests of the allgather protocol
  py::class_<FileBootstrap>(m, "FileBootstrap")
      .def(py::init<std::string, int, int>(), py::arg("dir"), py::arg("rank"), py::arg("world"))
      .def("allgather", [](FileBootstrap &b, const std::string &phase, py::bytes payload) {
        std::vector<py::bytes> out;
        for (auto &s : b.allgather(phase, std::string(payload))) out.emplace_back(s);
        return out;
      });

  // native RCCL wire (csrc/src/wire.hip): torch-free cross-rank transport
  py::class_<RcclWire>(m, "RcclWire")
      .def(py::init([](int device, int rank, int world, py::bytes uid) {
             return new RcclWire(device, rank, world, std::string(uid));
           }),
           py::arg("device"), py::arg("rank"), py::arg("world"), py::arg("uid"))
      .def_static("unique_id", []() { return py::bytes(RcclWire::unique_id()); })
      .def("add_send", &RcclWire::add_send, py::arg("group"), py::arg("ptr"), py::arg("bytes"),
           py::arg("peer"), py::arg("tag"))
      .def("add_recv", &RcclWire::add_recv, py::arg("group"), py::arg("ptr"), py::arg("bytes"),
           py::arg("peer"), py::arg("tag"))
      .def("finalize", &RcclWire::finalize)
      .def("post", &RcclWire::post, py::arg("group"), py::arg("stream"))
      .def("barrier", &RcclWire::barrier, py::arg("stream"))
      .def("rank", &RcclWire::rank)
      .def("world", &RcclWire::world)
      .def("device", &RcclWire::device);

  // topology utilities (csrc/src/topo.hip)
  m.def("gpu_distance", &gpu_distance);
  m.def("peer_copy_bandwidth", &peer_copy_bandwidth, py::arg("src"), py::arg("dst"),
        py::arg("bytes"), py::arg("iters") = 10);
  py::class_<GpuInfo>(m, "GpuInfo")
      .def_readonly("name", &GpuInfo::name)
      .def_readonly("pci", &GpuInfo::pci)
      .def_readonly("total_mem", &GpuInfo::totalMem)
      .def_readonly("cu_count", &GpuInfo::cuCount);
  m.def("gpu_info", &gpu_info);

  m.def("device_mem_info", [](int dev) {
    STENCIL_HIP(hipSetDevice(dev));
    size_t freeB = 0, totalB = 0;
    STENCIL_HIP(hipMemGetInfo(&freeB, &totalB));
    return py::make_tuple((int64_t)freeB, (int64_t)totalB);
  });
  m.def("device_count", []() {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
  });
  m.def("device_synchronize_all", []() {
    int n = 0;
    STENCIL_HIP(hipGetDeviceCount(&n));
    for (int i = 0; i < n; ++i) {
      STENCIL_HIP(hipSetDevice(i));
      STENCIL_HIP(hipDeviceSynchronize());
    }
  });
}
