// Serial CPU engine: the correctness oracle for the HIP engine and the
// no-GPU plumbing path (BASELINE config 1).
#include "engine.h"
#include "walk.h"

#include <atomic>
#include <cmath>
#include <cstring>
#include <stdexcept>
#include <thread>

namespace pumitally {

int default_max_steps(const Mesh &m) {
  const int c = (int)std::ceil(std::cbrt((double)m.nelems));
  return std::max(1000, 32 * c + 64);
}

std::vector<double> normalize_flux(const Mesh &m, const std::vector<double> &flux) {
  if ((int64_t)flux.size() != m.nelems)
    throw std::runtime_error(
        "normalize_flux: flux size != nelems (for grouped tallies normalize "
        "each group slice)");
  std::vector<double> out(m.nelems);
  for (int64_t e = 0; e < m.nelems; ++e) out[e] = flux[e] / m.volumes[e];
  return out;
}

namespace {

class CpuEngine final : public Engine {
public:
  CpuEngine(Mesh mesh, int64_t n, int groups, int scores)
      : mesh_(std::move(mesh)), n_(n) {
    ngroups = groups < 1 ? 1 : groups;
    nscores = scores < 1 ? 1 : scores;
    flux_.assign(mesh_.nelems * ngroups * nscores, 0.0);
    pos_.resize(n_ * 3);
    elem_.assign(n_, 0);
    escaped_.assign(n_, 0);
    const Vec3 c0 = mesh_.nelems > 0 ? mesh_.centroid(0) : Vec3{0, 0, 0};
    for (int64_t i = 0; i < n_; ++i) {
      pos_[i * 3] = c0.x;
      pos_[i * 3 + 1] = c0.y;
      pos_[i * 3 + 2] = c0.z;
    }
    loc_tol_ = loc_tol_rel() * norm(mesh_.bbox_hi - mesh_.bbox_lo);
    walk_fp32 = default_walk_fp32();
    reflective = default_reflective();
  }

  int64_t num_particles() const override { return n_; }
  const Mesh &mesh() const override { return mesh_; }

  void copy_initial_position(const double *p, int64_t n) override {
    check_n(n);
    auto locate_range = [&](int64_t lo, int64_t hi) {
      int64_t my_loose = 0;
      for (int64_t i = lo; i < hi; ++i) {
        const Vec3 q{p[i * 3], p[i * 3 + 1], p[i * 3 + 2]};
        bool loose = false;
        elem_[i] = mesh_.locate(q, loc_tol_, &loose);
        my_loose += loose;
        pos_[i * 3] = q.x;
        pos_[i * 3 + 1] = q.y;
        pos_[i * 3 + 2] = q.z;
        escaped_[i] = 0;
      }
      loose_ += my_loose;
    };
    const unsigned hw = std::thread::hardware_concurrency();
    if (n >= 65536 && hw > 1) {
      const int nthreads = (int)std::min<unsigned>(hw, 64);
      const int64_t per = (n + nthreads - 1) / nthreads;
      std::vector<std::thread> workers;
      for (int t = 0; t < nthreads; ++t)
        workers.emplace_back([&, t] {
          locate_range(t * per, std::min<int64_t>(n, (t + 1) * per));
        });
      for (auto &w : workers) w.join();
      return;
    }
    locate_range(0, n);
  }

  void move(const double *origin, const double *dest, const int8_t *flying,
            const double *weights, int64_t n, const uint16_t *groups = nullptr,
            const double *responses = nullptr) override {
    check_n(n);
    const int steps = max_steps > 0 ? max_steps : default_max_steps(mesh_);
    // Thread-parallel over particles for large batches (the reference's CPU
    // path is Kokkos OpenMP).  Each thread tallies into a private flux
    // array; partials are summed in thread-index order, so results are
    // DETERMINISTIC for a fixed thread count (and exactly serial-order for
    // one thread).  Small batches stay serial for bitwise stability of the
    // golden tests.
    unsigned hw = std::thread::hardware_concurrency();
    if (const char *env = getenv("PUMITALLY_CPU_THREADS")) {
      const int v = atoi(env);
      hw = v > 0 ? (unsigned)v : 1;
    }
    const int nthreads =
        (n >= 65536 && hw > 1) ? (int)std::min<unsigned>(hw, 64) : 1;
    if (nthreads > 1) {
      // full tally shape INCLUDING the score dimension: the scored add
      // writes k*ngroups*nelems + group*nelems + elem for k < nscores
      // (an nelems*ngroups-sized partial overflows the heap -- found by
      // tools/part_world2_soak at 400k particles with nscores=2)
      std::vector<std::vector<double>> partial(
          nthreads, std::vector<double>(flux_.size(), 0.0));
      std::atomic<int64_t> lost{0}, reloc{0};
      std::vector<std::thread> workers;
      const int64_t per = (n + nthreads - 1) / nthreads;
      for (int t = 0; t < nthreads; ++t) {
        workers.emplace_back([&, t] {
          const int64_t lo = t * per, hi = std::min<int64_t>(n, lo + per);
          int64_t my_lost = 0, my_reloc = 0;
          for (int64_t i = lo; i < hi; ++i)
            move_one(origin, dest, flying, weights, groups, responses, i,
                     steps, partial[t].data(), my_lost, my_reloc);
          lost += my_lost;
          reloc += my_reloc;
        });
      }
      for (auto &w : workers) w.join();
      for (int t = 0; t < nthreads; ++t)
        for (size_t e = 0; e < flux_.size(); ++e)
          flux_[e] += partial[t][e];
      stats_.lost_particles += lost.load();
      stats_.relocated += reloc.load();
      stats_.moves++;
      return;
    }
    int64_t lost = 0, reloc = 0;
    for (int64_t i = 0; i < n; ++i)
      move_one(origin, dest, flying, weights, groups, responses, i, steps,
               flux_.data(), lost, reloc);
    stats_.lost_particles += lost;
    stats_.relocated += reloc;
    stats_.moves++;
  }

  // One particle of a move(): phase A (relocation, skipped for escaped
  // particles -- behavioral pin, see engine.h) + phase B tallied walk.
  void move_one(const double *origin, const double *dest,
                const int8_t *flying, const double *weights,
                const uint16_t *groups, const double *responses, int64_t i,
                int steps, double *flux_out, int64_t &lost, int64_t &reloc) {
    if (!flying[i]) return;
    Vec3 o{pos_[i * 3], pos_[i * 3 + 1], pos_[i * 3 + 2]};
    if (origin && !escaped_[i]) {
      const Vec3 q{origin[i * 3], origin[i * 3 + 1], origin[i * 3 + 2]};
      if (q.x != o.x || q.y != o.y || q.z != o.z) {
        bool loose = false;
        elem_[i] = mesh_.locate(q, loc_tol_, &loose);
        if (loose) loose_++; // rare; atomic is fine in the threaded path
        o = q;
        reloc++;
      }
    }
    if (elem_[i] < 0) {
      // outside the mesh: nothing to tally; remember requested position
      pos_[i * 3] = o.x;
      pos_[i * 3 + 1] = o.y;
      pos_[i * 3 + 2] = o.z;
      return;
    }
    const Vec3 d{dest[i * 3], dest[i * 3 + 1], dest[i * 3 + 2]};
    int32_t out_elem;
    Vec3 out_pos;
    bool out_esc;
    const int64_t goff =
        groups ? (int64_t)(groups[i] % ngroups) * mesh_.nelems : 0;
    const int64_t gsz = (int64_t)ngroups * mesh_.nelems;
    const double *resp = responses ? responses + i * nscores : nullptr;
    auto add = [&](int32_t e, double v) {
      if (!resp) {
        flux_out[goff + e] += v;
        return;
      }
      for (int k = 0; k < nscores; ++k)
        flux_out[k * gsz + goff + e] += v * resp[k];
    };
    const uint32_t *bc =
        mesh_.face_bc_bits.empty() ? nullptr : mesh_.face_bc_bits.data();
    const int32_t *pix =
        mesh_.periodic_idx.empty() ? nullptr : mesh_.periodic_idx.data();
    if (walk_fp32)
      walk_segment32<true>(mesh_.planes.data(), mesh_.planes32.data(),
                           mesh_.nbr.data(), elem_[i], o, d, weights[i],
                           steps, add, &out_elem, &out_pos, &out_esc,
                           reflective, bc, pix, mesh_.periodic_elem.data(),
                           mesh_.periodic_shift.data());
    else
      walk_segment<true>(mesh_.planes.data(), mesh_.nbr.data(), elem_[i], o,
                         d, weights[i], steps, add, &out_elem, &out_pos,
                         &out_esc, reflective, bc, pix,
                         mesh_.periodic_elem.data(),
                         mesh_.periodic_shift.data());
    if (out_elem == kWalkLost) {
      lost++;
      record_lost(i, out_pos);
      out_elem = elem_[i];
    }
    elem_[i] = out_elem;
    pos_[i * 3] = out_pos.x;
    pos_[i * 3 + 1] = out_pos.y;
    pos_[i * 3 + 2] = out_pos.z;
    escaped_[i] = out_esc ? 1 : 0;
  }

  void walk_raw(int64_t n, const double *pos, const double *dest,
                const int32_t *elem, const double *weights, double *out_pos,
                int32_t *out_elem, int8_t *out_status,
                const uint16_t *groups = nullptr,
                const double *responses = nullptr,
                double *out_dest = nullptr,
                const double *in_t = nullptr,
                const int32_t *in_prev = nullptr, double *out_o = nullptr,
                double *out_t = nullptr,
                int32_t *out_prev = nullptr) override {
    const int steps = max_steps > 0 ? max_steps : default_max_steps(mesh_);
    const unsigned hw = std::thread::hardware_concurrency();
    if (n >= 65536 && hw > 1) {
      const int nthreads = (int)std::min<unsigned>(hw, 64);
      std::vector<std::vector<double>> partial(
          nthreads, std::vector<double>(flux_.size(), 0.0));
      std::atomic<int64_t> lost{0};
      std::vector<std::thread> workers;
      const int64_t per = (n + nthreads - 1) / nthreads;
      for (int t = 0; t < nthreads; ++t) {
        workers.emplace_back([&, t] {
          const int64_t lo = t * per, hi = std::min<int64_t>(n, lo + per);
          int64_t my_lost = 0;
          for (int64_t i = lo; i < hi; ++i)
            walk_raw_one(pos, dest, elem, weights, groups, responses, out_pos,
                         out_elem, out_status, out_dest, in_t, in_prev,
                         out_o, out_t, out_prev, i, steps,
                         partial[t].data(), my_lost);
          lost += my_lost;
        });
      }
      for (auto &w : workers) w.join();
      for (int t = 0; t < nthreads; ++t)
        for (size_t e = 0; e < flux_.size(); ++e) flux_[e] += partial[t][e];
      stats_.lost_particles += lost.load();
      return;
    }
    int64_t lost = 0;
    for (int64_t i = 0; i < n; ++i)
      walk_raw_one(pos, dest, elem, weights, groups, responses, out_pos,
                   out_elem, out_status, out_dest, in_t, in_prev, out_o,
                   out_t, out_prev, i, steps, flux_.data(), lost);
    stats_.lost_particles += lost;
  }

  void walk_raw_one(const double *pos, const double *dest,
                    const int32_t *elem, const double *weights,
                    const uint16_t *groups, const double *responses,
                    double *out_pos, int32_t *out_elem, int8_t *out_status,
                    double *out_dest, const double *in_t,
                    const int32_t *in_prev, double *out_o, double *out_t,
                    int32_t *out_prev, int64_t i, int steps,
                    double *flux_out, int64_t &lost) {
    {
      const Vec3 o{pos[i * 3], pos[i * 3 + 1], pos[i * 3 + 2]};
      const Vec3 d{dest[i * 3], dest[i * 3 + 1], dest[i * 3 + 2]};
      int32_t oe;
      Vec3 op;
      bool esc;
      const int64_t goff =
          groups ? (int64_t)(groups[i] % ngroups) * mesh_.nelems : 0;
      const int64_t gsz = (int64_t)ngroups * mesh_.nelems;
      const double *resp = responses ? responses + i * nscores : nullptr;
      auto add = [&](int32_t e, double v) {
        if (!resp) {
          flux_out[goff + e] += v;
          return;
        }
        for (int k = 0; k < nscores; ++k)
          flux_out[k * gsz + goff + e] += v * resp[k];
      };
      const uint32_t *bc =
          mesh_.face_bc_bits.empty() ? nullptr : mesh_.face_bc_bits.data();
      const int32_t *pix =
          mesh_.periodic_idx.empty() ? nullptr : mesh_.periodic_idx.data();
      Vec3 od{d.x, d.y, d.z};
      const double rt = in_t ? in_t[i] : 0.0;
      const int32_t rp = in_prev ? in_prev[i] : -1;
      Vec3 oo{o.x, o.y, o.z};
      double ot = 0.0;
      int32_t opv = -1;
      if (walk_fp32)
        walk_segment32<true>(mesh_.planes.data(), mesh_.planes32.data(),
                             mesh_.nbr.data(), elem[i], o, d, weights[i],
                             steps, add, &oe, &op, &esc, reflective, bc, pix,
                             mesh_.periodic_elem.data(),
                             mesh_.periodic_shift.data(), &od, rt, rp, &oo,
                             &ot, &opv);
      else
        walk_segment<true>(mesh_.planes.data(), mesh_.nbr.data(), elem[i], o,
                           d, weights[i], steps, add, &oe, &op, &esc,
                           reflective, bc, pix, mesh_.periodic_elem.data(),
                           mesh_.periodic_shift.data(), &od, rt, rp, &oo,
                           &ot, &opv);
      int8_t st = 0;
      if (oe == kWalkLost) {
        st = 3;
        oe = elem[i];
        lost++;
        record_lost(i, op);
      } else if (esc) {
        st = 1;
      } else if (oe < -1) {
        st = 2;
      }
      out_elem[i] = oe;
      out_pos[i * 3] = op.x;
      out_pos[i * 3 + 1] = op.y;
      out_pos[i * 3 + 2] = op.z;
      out_status[i] = st;
      if (out_dest) {
        out_dest[i * 3] = od.x;
        out_dest[i * 3 + 1] = od.y;
        out_dest[i * 3 + 2] = od.z;
      }
      if (out_o) {
        out_o[i * 3] = oo.x;
        out_o[i * 3 + 1] = oo.y;
        out_o[i * 3 + 2] = oo.z;
      }
      if (out_t) out_t[i] = ot;
      if (out_prev) out_prev[i] = opv;
    }
  }

  void end_batch() override {
    if (batch_sum_.empty()) {
      batch_sum_.assign(flux_.size(), 0.0);
      batch_sq_.assign(flux_.size(), 0.0);
    }
    for (size_t e = 0; e < flux_.size(); ++e) {
      batch_sum_[e] += flux_[e];
      batch_sq_[e] += flux_[e] * flux_[e];
      flux_[e] = 0.0;
    }
    nbatches_++;
  }
  std::vector<double> batch_sum() const override {
    return batch_sum_.empty() ? std::vector<double>(flux_.size(), 0.0) : batch_sum_;
  }
  std::vector<double> batch_sum_sq() const override {
    return batch_sq_.empty() ? std::vector<double>(flux_.size(), 0.0) : batch_sq_;
  }
  int64_t num_batches() const override { return nbatches_; }

  std::vector<double> flux() const override { return flux_; }
  std::vector<int32_t> elem_ids() const override { return elem_; }
  std::vector<double> positions() const override { return pos_; }
  std::vector<uint8_t> escaped() const override { return escaped_; }
  const EngineStats &stats() const override {
    stats_.loose_localizations = loose_.load();
    return stats_;
  }

  std::vector<double> lost_records() const override {
    const int64_t k =
        std::min<int64_t>(lost_rec_n_.load(), kMaxLostRecords);
    return {lost_rec_.begin(), lost_rec_.begin() + k * 4};
  }

  void set_flux(const double *f, int64_t ne) override {
    if (ne != (int64_t)flux_.size()) throw std::runtime_error("set_flux size mismatch");
    std::memcpy(flux_.data(), f, ne * sizeof(double));
  }

  void set_particle_state(const double *pos, const int32_t *elem,
                          const uint8_t *escaped, int64_t n) override {
    check_n(n);
    std::memcpy(pos_.data(), pos, n * 3 * sizeof(double));
    std::memcpy(elem_.data(), elem, n * sizeof(int32_t));
    std::memcpy(escaped_.data(), escaped, n);
  }

private:
  void check_n(int64_t n) const {
    if (n != n_) throw std::runtime_error("particle count mismatch");
  }

  void record_lost(int64_t i, Vec3 p) {
    const int64_t k = lost_rec_n_.fetch_add(1);
    if (k < kMaxLostRecords) {
      lost_rec_[k * 4] = (double)i;
      lost_rec_[k * 4 + 1] = p.x;
      lost_rec_[k * 4 + 2] = p.y;
      lost_rec_[k * 4 + 3] = p.z;
    }
  }

  Mesh mesh_;
  int64_t n_;
  double loc_tol_;
  std::vector<double> flux_, pos_, batch_sum_, batch_sq_;
  int64_t nbatches_ = 0;
  std::vector<int32_t> elem_;
  std::vector<uint8_t> escaped_;
  mutable EngineStats stats_;
  std::atomic<int64_t> loose_{0};
  std::vector<double> lost_rec_ =
      std::vector<double>((size_t)kMaxLostRecords * 4, 0.0);
  std::atomic<int64_t> lost_rec_n_{0};
};

} // namespace

std::unique_ptr<Engine> make_cpu_engine(Mesh mesh, int64_t num_particles,
                                        int ngroups, int nscores) {
  return std::make_unique<CpuEngine>(std::move(mesh), num_particles, ngroups,
                                     nscores);
}

} // namespace pumitally
