// DistributedDomain (C++) implementation. See distributed.hpp for the
// design; plans and wire formats are identical to the Python orchestrator
// (planning.hpp == planning.py, pinned by tests/test_native_plan.py).
#include "stencil_amd/distributed.hpp"
#include "stencil_amd/hip_check.hpp"
#include "stencil_amd/log.hpp"

#include <hip/hip_runtime.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <fstream>
#include <sstream>
#include <stdexcept>
#include <thread>
#include <unistd.h>

namespace stencil_amd {

//// FileBootstrap ////

FileBootstrap::FileBootstrap(std::string dir, int rank, int world)
    : dir_(std::move(dir)), rank_(rank), world_(world) {
  if (dir_.empty()) throw std::runtime_error("FileBootstrap: empty dir");
}

std::vector<std::string> FileBootstrap::allgather(const std::string &phase,
                                                 const std::string &payload) {
  // write <dir>/<phase>.<seq>.<rank> atomically (tmp + rename), then
  // poll-read every rank's file. The per-call sequence number keeps
  // repeated collectives (or two DistributedDomains constructed in the
  // same order on every rank) from matching stale files; the bootstrap
  // dir itself must be fresh per job (run_native_mp.sh uses mktemp -d).
  const std::string base = dir_ + "/" + phase + "." + std::to_string(seq_++) + ".";
  const std::string tmp = base + std::to_string(rank_) + ".tmp";
  const std::string fin = base + std::to_string(rank_);
  {
    std::ofstream f(tmp, std::ios::binary);
    if (!f) throw std::runtime_error("FileBootstrap: cannot write " + tmp);
    f.write(payload.data(), (std::streamsize)payload.size());
  }
  if (std::rename(tmp.c_str(), fin.c_str()) != 0)
    throw std::runtime_error("FileBootstrap: rename failed for " + fin);

  std::vector<std::string> out(world_);
  const auto deadline = std::chrono::steady_clock::now() + std::chrono::seconds(120);
  for (int r = 0; r < world_; ++r) {
    const std::string path = base + std::to_string(r);
    for (;;) {
      std::ifstream f(path, std::ios::binary);
      if (f) {
        std::ostringstream ss;
        ss << f.rdbuf();
        out[r] = ss.str();
        break;
      }
      if (std::chrono::steady_clock::now() > deadline)
        throw std::runtime_error("FileBootstrap: timeout waiting for " + path);
      std::this_thread::sleep_for(std::chrono::milliseconds(5));
    }
  }
  return out;
}

//// DistributedDomain ////

namespace {
// disambiguates the overloaded LocalDomain::halo_extent static
Vec3 halo_extent_of(const Vec3 &d, const Vec3 &sz, const Radius &r) {
  return LocalDomain::halo_extent(d, sz, r);
}
int env_int(const char *name, const char *alt, int dflt) {
  if (const char *e = getenv(name)) return atoi(e);
  if (alt)
    if (const char *e = getenv(alt)) return atoi(e);
  return dflt;
}
} // namespace

DistributedDomain::DistributedDomain(int64_t x, int64_t y, int64_t z) : size_(x, y, z) {
  rank_ = env_int("STENCIL_RANK", "RANK", 0);
  world_ = env_int("STENCIL_WORLD", "WORLD_SIZE", 1);
  if (world_ > 1) {
    const char *dir = getenv("STENCIL_BOOTSTRAP_DIR");
    if (!dir)
      throw std::runtime_error(
          "world > 1 needs STENCIL_BOOTSTRAP_DIR (shared dir for the control plane)");
    boot_ = std::make_unique<FileBootstrap>(dir, rank_, world_);
  }
}

int64_t DistributedDomain::add_data(int64_t elemSize, const std::string &name) {
  data_.emplace_back(elemSize, name);
  return (int64_t)data_.size() - 1;
}

void DistributedDomain::set_exchange_groups(const std::vector<std::vector<int64_t>> &groups) {
  // 3 engine launch slots per group x kGroups=12 (engine.hpp); RcclWire
  // mirrors the same bound
  if (groups.size() > 4) throw std::runtime_error("at most 4 exchange groups");
  groups_ = groups;
  for (auto &g : groups_) std::sort(g.begin(), g.end());
}

void DistributedDomain::gather_slots_(std::vector<Slot> &slots) {
  if (gpus_.empty()) {
    int n = 0;
    STENCIL_HIP(hipGetDeviceCount(&n));
    if (n == 0) throw std::runtime_error("no HIP devices visible");
    gpus_ = {world_ == 1 ? 0 : env_int("STENCIL_LOCAL_RANK", "LOCAL_RANK", rank_) % n};
  }
  char host[256] = {0};
  gethostname(host, sizeof(host) - 1);
  std::string payload = std::string(host) + "\n";
  for (size_t i = 0; i < gpus_.size(); ++i)
    payload += (i ? "," : "") + std::to_string(gpus_[i]);
  std::vector<std::string> infos;
  if (world_ == 1) {
    infos = {payload};
  } else {
    infos = boot_->allgather("slots", payload);
  }
  // node index by first appearance of the hostname (matches Comm.node_of_rank)
  std::vector<std::string> order;
  for (int r = 0; r < world_; ++r) {
    const std::string h = infos[r].substr(0, infos[r].find('\n'));
    if (std::find(order.begin(), order.end(), h) == order.end()) order.push_back(h);
  }
  for (int r = 0; r < world_; ++r) {
    const size_t nl = infos[r].find('\n');
    const std::string h = infos[r].substr(0, nl);
    const int node = (int)(std::find(order.begin(), order.end(), h) - order.begin());
    std::stringstream ss(infos[r].substr(nl + 1));
    std::string tok;
    int li = 0;
    while (std::getline(ss, tok, ','))
      slots.push_back({r, li++, atoi(tok.c_str()), node});
  }
}

void DistributedDomain::realize() {
  if (realized_) throw std::runtime_error("realize() called twice");
  auto now = []() { return std::chrono::steady_clock::now(); };
  auto secs = [](auto a, auto b) { return std::chrono::duration<double>(b - a).count(); };
  auto t0 = now();
  std::vector<Slot> slots;
  gather_slots_(slots);
  setupTimes_["topo"] = secs(t0, now());
  t0 = now();

  switch (strategy_) {
  case PlacementStrategy::Trivial:
    placement_ = std::make_unique<TrivialPlacement>(size_, radius_, slots);
    break;
  case PlacementStrategy::IntraNodeRandom:
    placement_ = std::make_unique<IntraNodeRandomPlacement>(size_, radius_, slots);
    break;
  case PlacementStrategy::NodeAware:
  default:
    placement_ = std::make_unique<NodeAwarePlacement>(size_, radius_, slots, &halo_extent_of);
    break;
  }

  setupTimes_["placement"] = secs(t0, now());
  t0 = now();
  const int nLocal = placement_->num_local(rank_);
  for (int li = 0; li < nLocal; ++li) {
    const Vec3 idx = placement_->get_idx(rank_, li);
    auto d = std::make_shared<LocalDomain>(placement_->subdomain_size(idx),
                                           placement_->subdomain_origin(idx),
                                           placement_->get_cuda(idx));
    d->set_radius(radius_);
    for (auto &q : data_) d->add_data(q.first, q.second);
    d->realize();
    domains_.push_back(std::move(d));
  }
  LOG_INFO("realize: rank %d/%d, %d local domain(s), dim (%lld,%lld,%lld)", rank_, world_,
           nLocal, (long long)placement_->dim().x, (long long)placement_->dim().y,
           (long long)placement_->dim().z);
  engine_ = std::make_unique<ExchangeEngine>(domains_);
  {
    std::set<int> devs;
    for (auto &d : domains_) devs.insert(d->gpu());
    if (devs.size() > 1) engine_->enable_peer_all();
  }

  if (groups_.empty()) {
    groups_.emplace_back();
    for (int64_t qi = 0; qi < (int64_t)data_.size(); ++qi) groups_.back().push_back(qi);
  }
  hasWire_.assign(groups_.size(), false);
  setupTimes_["realize"] = secs(t0, now());
  t0 = now();

  const ExchangePlan plan = plan_exchange(*placement_, radius_, rank_, &halo_extent_of);
  setupTimes_["plan"] = secs(t0, now());
  t0 = now();
  std::vector<int64_t> elemSizes;
  int64_t esTotal = 0;
  for (auto &q : data_) {
    elemSizes.push_back(q.first);
    esTotal += q.first;
  }

  // same-rank direct-write translate jobs (one per group x region)
  for (size_t g = 0; g < groups_.size(); ++g)
    for (const TranslatePlanItem &t : plan.translates) {
      LocalDomain &src = *domains_[t.srcLocal];
      LocalDomain &dst = *domains_[t.dstLocal];
      engine_->add_translate(t.srcLocal, t.dstLocal, src.halo_pos(t.dir, false),
                             dst.halo_pos(-t.dir, true), t.ext, (int)g, groups_[g]);
      if (g == 0) bytesTranslate_ += t.ext.flatten() * esTotal;
    }

  // cross-rank packed wire (RcclWire)
  const auto seq = pair_seq_tags(plan);
  const int ng = (int)groups_.size();
  struct WireOp {
    int64_t buf;
    int peer;
    int64_t tag;
    bool send;
  };
  std::vector<std::vector<WireOp>> wireOps(ng);
  for (int g = 0; g < ng; ++g) {
    for (const auto *items : {&plan.sends, &plan.recvs}) {
      const bool isSend = items == &plan.sends;
      for (const WirePlanItem &item : *items) {
        std::vector<WireChunk> chunks;
        const int64_t total = wire_layout(item.messages, elemSizes, groups_[g], chunks);
        const int64_t buf = engine_->create_buffer(item.localId, total);
        LocalDomain &dom = *domains_[item.localId];
        for (const WireChunk &c : chunks) {
          const PlanMessage &m = item.messages[c.msgIndex];
          if (isSend) {
            engine_->add_pack(item.localId, buf, c.offset, dom.halo_pos(m.dir, false), m.ext,
                              c.qi, 3 * g);
          } else {
            engine_->add_unpack(item.localId, buf, c.offset, dom.halo_pos(-m.dir, true), m.ext,
                                c.qi, 3 * g);
          }
        }
        const int64_t tag = seq.at({item.peerRank, item.srcGid, item.dstGid}) * ng + g;
        wireOps[g].push_back({buf, item.peerRank, tag, isSend});
        hasWire_[g] = true;
        if (isSend && g == 0)
          for (const PlanMessage &m : item.messages) bytesWire_ += m.volume() * esTotal;
      }
    }
  }
  engine_->finalize();

  bool anyWire = false;
  for (bool h : hasWire_) anyWire = anyWire || h;
  if (world_ > 1) {
    // every rank must agree (collective comm creation); vote via bootstrap
    const auto votes = boot_->allgather("wire_vote", anyWire ? "1" : "0");
    bool someWire = false;
    for (auto &v : votes) someWire = someWire || (v == "1");
    if (someWire) {
      std::set<int> devs;
      for (auto &g : wireOps)
        for (auto &op : g) devs.insert(engine_->buffer_device(op.buf));
      if (devs.size() > 1)
        throw std::runtime_error("C++ RCCL wire drives one device per rank; "
                                 "use one process per GPU for cross-rank runs");
      wireDev_ = devs.empty() ? domains_[0]->gpu() : *devs.begin();
      const std::string uid = rank_ == 0 ? RcclWire::unique_id() : std::string();
      const auto uids = boot_->allgather("rccl_uid", uid);
      LOG_INFO("wire: RCCL communicator on device %d (rank %d/%d)", wireDev_, rank_, world_);
      wire_ = std::make_unique<RcclWire>(wireDev_, rank_, world_, uids[0]);
      for (int g = 0; g < ng; ++g)
        for (const WireOp &op : wireOps[g]) {
          if (op.send)
            wire_->add_send(g, engine_->buffer_ptr(op.buf), engine_->buffer_bytes(op.buf),
                            op.peer, op.tag);
          else
            wire_->add_recv(g, engine_->buffer_ptr(op.buf), engine_->buffer_bytes(op.buf),
                            op.peer, op.tag);
        }
      wire_->finalize();
    }
  } else if (anyWire) {
    throw std::runtime_error("cross-rank messages planned at world=1");
  }
  setupTimes_["create"] = secs(t0, now());
  // plan files (reference src/stencil.cu:482-637; Python _write_plan_files
  // writes the same shape)
  if (const char *prefix = getenv("STENCIL_OUTPUT_PREFIX")) {
    int64_t esTotal2 = 0;
    for (auto &q : data_) esTotal2 += q.first;
    FILE *f = fopen((std::string(prefix) + "plan_" + std::to_string(rank_) + ".txt").c_str(), "w");
    if (f) {
      const Vec3 dim = placement_->dim();
      fprintf(f, "rank %d world %d dim (%lld, %lld, %lld)\n", rank_, world_, (long long)dim.x,
              (long long)dim.y, (long long)dim.z);
      for (const TranslatePlanItem &t : plan.translates)
        fprintf(f, "direct_kernel dir=(%lld, %lld, %lld) src_local=%d dst_local=%d bytes=%lld\n",
                (long long)t.dir.x, (long long)t.dir.y, (long long)t.dir.z, t.srcLocal,
                t.dstLocal, (long long)(t.ext.flatten() * esTotal2));
      for (const WirePlanItem &it : plan.sends)
        for (const PlanMessage &m : it.messages)
          fprintf(f, "rccl_send dir=(%lld, %lld, %lld) dst_rank=%d src_gid=%lld dst_gid=%lld bytes=%lld\n",
                  (long long)m.dir.x, (long long)m.dir.y, (long long)m.dir.z, it.peerRank,
                  (long long)m.srcGid, (long long)m.dstGid, (long long)(m.volume() * esTotal2));
      for (const WirePlanItem &it : plan.recvs)
        for (const PlanMessage &m : it.messages)
          fprintf(f, "rccl_recv dir=(%lld, %lld, %lld) src_rank=%d src_gid=%lld dst_gid=%lld bytes=%lld\n",
                  (long long)m.dir.x, (long long)m.dir.y, (long long)m.dir.z, it.peerRank,
                  (long long)m.srcGid, (long long)m.dstGid, (long long)(m.volume() * esTotal2));
      fclose(f);
    }
  }
  realized_ = true;
}

void DistributedDomain::exchange_begin(int group) {
  engine_->launch_translates(group);
  if (hasWire_[group]) engine_->launch_packs(3 * group);
}

void DistributedDomain::exchange_end(int group) {
  if (hasWire_[group] && wire_) {
    // stream-ordered: packs already sit on the pack stream, the grouped
    // send/recv posts behind them, unpacks enqueue behind the recvs
    wire_->post(group, engine_->pack_stream_handle(wireDev_));
    engine_->launch_unpacks(3 * group);
  }
  engine_->sync_all();
}

void DistributedDomain::exchange(int group) {
  exchange_begin(group);
  exchange_end(group);
}

void DistributedDomain::swap() {
  for (auto &d : domains_) d->swap();
}

void DistributedDomain::write_paraview(const std::string &prefix) {
  for (int li = 0; li < num_local(); ++li) {
    LocalDomain &d = *domains_[li];
    const Rect3 r = local_rect(li);
    const Vec3 ext = r.extent();
    const Rect3 full = d.full_region();
    const Vec3 pos = r.lo - full.lo; // allocation coords
    const int64_t nq = (int64_t)data_.size();
    std::vector<std::vector<double>> vals(nq);
    for (int64_t qi = 0; qi < nq; ++qi) {
      const int64_t es = data_[qi].first;
      std::vector<char> raw(ext.flatten() * es);
      d.region_to_host(raw.data(), pos, ext, qi, false);
      vals[qi].resize(ext.flatten());
      for (int64_t i = 0; i < ext.flatten(); ++i) {
        switch (es) { // size-canonical interpretation (fp32/fp64/int16/u8)
        case 8: vals[qi][i] = ((const double *)raw.data())[i]; break;
        case 4: vals[qi][i] = ((const float *)raw.data())[i]; break;
        case 2: vals[qi][i] = ((const int16_t *)raw.data())[i]; break;
        default: vals[qi][i] = ((const uint8_t *)raw.data())[i]; break;
        }
      }
    }
    const Vec3 idx = placement_->get_idx(rank_, li);
    const int64_t gid = placement_->linearize(idx);
    FILE *f = fopen((prefix + std::to_string(gid) + ".txt").c_str(), "w");
    if (!f) throw std::runtime_error("write_paraview: cannot open output");
    fprintf(f, "Z,Y,X");
    for (int64_t qi = 0; qi < nq; ++qi) {
      const std::string name =
          data_[qi].second.empty() ? "q" + std::to_string(qi) : data_[qi].second;
      fprintf(f, ",%s", name.c_str());
    }
    fprintf(f, "\n");
    int64_t i = 0;
    for (int64_t z = r.lo.z; z < r.hi.z; ++z)
      for (int64_t y = r.lo.y; y < r.hi.y; ++y)
        for (int64_t x = r.lo.x; x < r.hi.x; ++x, ++i) {
          fprintf(f, "%lld,%lld,%lld", (long long)z, (long long)y, (long long)x);
          for (int64_t qi = 0; qi < nq; ++qi) fprintf(f, ",%.17g", vals[qi][i]);
          fprintf(f, "\n");
        }
    fclose(f);
  }
}

Rect3 DistributedDomain::local_rect(int li) const {
  const Vec3 idx = placement_->get_idx(rank_, li);
  const Vec3 o = placement_->subdomain_origin(idx);
  return Rect3(o, o + placement_->subdomain_size(idx));
}

std::vector<Rect3> DistributedDomain::get_interior() const {
  // mirror core.py get_interior (reference src/stencil.cu:878-923)
  Vec3 shrinkLo(0, 0, 0), shrinkHi(0, 0, 0);
  for (int a = -1; a <= 1; ++a)
    for (int b = -1; b <= 1; ++b) {
      shrinkLo.x = std::max(shrinkLo.x, radius_.dir(-1, a, b));
      shrinkHi.x = std::max(shrinkHi.x, radius_.dir(1, a, b));
      shrinkLo.y = std::max(shrinkLo.y, radius_.dir(a, -1, b));
      shrinkHi.y = std::max(shrinkHi.y, radius_.dir(a, 1, b));
      shrinkLo.z = std::max(shrinkLo.z, radius_.dir(a, b, -1));
      shrinkHi.z = std::max(shrinkHi.z, radius_.dir(a, b, 1));
    }
  std::vector<Rect3> out;
  for (int li = 0; li < num_local(); ++li) {
    const Rect3 r = local_rect(li);
    Vec3 ilo, ihi;
    for (int i = 0; i < 3; ++i) {
      ilo[i] = std::min(r.lo[i] + shrinkLo[i], r.hi[i]);
      ihi[i] = std::max(r.hi[i] - shrinkHi[i], ilo[i]);
    }
    out.emplace_back(ilo, ihi);
  }
  return out;
}

std::vector<std::vector<Rect3>> DistributedDomain::get_exterior() const {
  // mirror core.py get_exterior (slide faces in, src/stencil.cu:927-977)
  const std::vector<Rect3> interiors = get_interior();
  std::vector<std::vector<Rect3>> out;
  for (int li = 0; li < num_local(); ++li) {
    const Rect3 r = local_rect(li);
    const Rect3 &in = interiors[li];
    std::vector<Rect3> boxes;
    Vec3 clo = r.lo, chi = r.hi;
    for (int axis = 0; axis < 3; ++axis) { // +x,+y,+z
      if (in.hi[axis] != chi[axis]) {
        Vec3 blo = clo;
        blo[axis] = in.hi[axis];
        boxes.emplace_back(blo, chi);
        chi[axis] = in.hi[axis];
      }
    }
    for (int axis = 0; axis < 3; ++axis) { // -x,-y,-z
      if (in.lo[axis] != clo[axis]) {
        Vec3 bhi = chi;
        bhi[axis] = in.lo[axis];
        boxes.emplace_back(clo, bhi);
        clo[axis] = in.lo[axis];
      }
    }
    out.push_back(std::move(boxes));
  }
  return out;
}

} // namespace stencil_amd
