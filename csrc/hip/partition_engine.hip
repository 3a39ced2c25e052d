// Stateful domain-decomposed tally engine (see partition_engine.h).
//
// GPU path: per-global-particle state arrays in HBM, device-compacted
// walk lists, the replicated engine's fused walk kernel via
// walk_raw_device, and 5-double handoff records [gid, pos x3, target
// global elem] exchanged over Comm::alltoallv_device (RCCL grouped
// send/recv over xGMI).  dest/weight/group are never shipped: every rank
// uploads the same global step arrays once and gathers by gid.
//
// CPU path: same algorithm with host loops (the differential oracle and
// the no-GPU fallback).
//
// Replaces the reference's pumipic picparts + migrate-inside-search
// (/root/reference/src/pumitally/PumiTallyImpl.cpp:111-145,433-459) with
// persistent residency: round-1's stateless Python driver re-localized
// and re-uploaded the whole batch every step (27x off the replicated
// engine); here a steady-state step uploads only dest/flying/weights.
#include "../comm/comm.h"
#include "../core/engine.h"
#include "../core/partition_engine.h"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstring>
#include <functional>
#include <memory>
#include <stdexcept>
#include <string>

namespace pumitally {

namespace {

#define PT_HIP_CHECK(expr)                                                     \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess)                                                      \
      throw std::runtime_error(std::string("HIP error at partition_engine:") + \
                               std::to_string(__LINE__) + ": " +               \
                               hipGetErrorString(_e));                         \
  } while (0)

// record: [gid, px, py, pz, target_gid, dx, dy, dz] -- the destination
// travels with the handoff because reflective/periodic restarts inside
// the walk MUTATE it (walk.h out_dest); the receiver's first continued
// walk uses the record's dest, while weights/groups/responses are still
// gathered by gid.
constexpr int kRecW = 8;
constexpr int kPBlock = 256;
constexpr int kMaxChunks = 8; // step() copy/walk pipeline depth

inline int pgrid(int64_t n) {
  int64_t b = (n + kPBlock - 1) / kPBlock;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

// Wave-aggregated append slot: one atomicAdd per 64-wide wavefront
// instead of one per lane (the single global counter was 1.9 ms/step of
// the prepare pass at 10M particles -- measured, profiles/README.md).
// Call from exactly the lanes that append; the ballot captures them.
__device__ __forceinline__ int64_t wave_append(unsigned long long *ctr) {
  const unsigned long long mask = __ballot(1);
  const int lane = (int)(threadIdx.x & 63u);
  const int leader = __ffsll((long long)mask) - 1;
  unsigned long long base = 0;
  if (lane == leader)
    base = atomicAdd(ctr, (unsigned long long)__popcll(mask));
  base = (unsigned long long)__shfl((long long)base, leader);
  return (int64_t)(base + __popcll(mask & ((1ull << lane) - 1ull)));
}

// ---------------------------------------------------------------------------
// kernels
// ---------------------------------------------------------------------------

__global__ void k_part_localize(const Plane *__restrict__ planes,
                                GridView grid,
                                const int32_t *__restrict__ lowner, int myrank,
                                const double *__restrict__ origins, int64_t n,
                                double tol, double *__restrict__ pos,
                                int32_t *__restrict__ elem,
                                uint8_t *__restrict__ res,
                                uint8_t *__restrict__ esc,
                                unsigned long long *__restrict__ claim,
                                unsigned long long *__restrict__ loose) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n; g += stride) {
    res[g] = 0;
    const Vec3 q{origins[g * 3], origins[g * 3 + 1], origins[g * 3 + 2]};
    bool lo = false;
    const int32_t le = grid_locate(grid, planes, q, tol, &lo);
    if (le >= 0 && lowner[le] == myrank) {
      if (lo) atomicAdd(loose, 1ull);
      res[g] = 1;
      esc[g] = 0;
      elem[g] = le;
      pos[g * 3] = q.x;
      pos[g * 3 + 1] = q.y;
      pos[g * 3 + 2] = q.z;
      atomicOr(&claim[g >> 6], 1ull << (g & 63));
    }
  }
}

__global__ void k_part_claim_rest(const unsigned long long *__restrict__ claim,
                                  const double *__restrict__ origins,
                                  int64_t n, double *__restrict__ pos,
                                  int32_t *__restrict__ elem,
                                  uint8_t *__restrict__ res,
                                  uint8_t *__restrict__ esc) {
  // rank 0 claims globally-unfound particles (outside the mesh): they
  // stay resident with elem = -1 and tally nothing (replicated-engine
  // semantics for out-of-mesh particles)
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n; g += stride) {
    if (!((claim[g >> 6] >> (g & 63)) & 1ull)) {
      res[g] = 1;
      esc[g] = 0;
      elem[g] = -1;
      pos[g * 3] = origins[g * 3];
      pos[g * 3 + 1] = origins[g * 3 + 1];
      pos[g * 3 + 2] = origins[g * 3 + 2];
    }
  }
}

// dep entry: 9 doubles [gid, px, py, pz, target_gid, owner, dx, dy, dz]
// Operates on the gid range [g_lo, g_hi): step() pipelines chunks of the
// batch so chunk c's prepare+walk overlap chunk c+1's H2D copies.  The
// chunk's walk list lives in list[g_lo ...] with its own counter
// (nwalk_ctr), so segments never collide.
__global__ void k_part_prepare(
    int64_t g_lo, int64_t g_hi, uint8_t *res /* read+clear, no restrict */,
    const uint8_t *__restrict__ esc, const int8_t *__restrict__ fly,
    const double *__restrict__ orig, double *__restrict__ pos,
    int32_t *__restrict__ elem, const Plane *__restrict__ planes,
    GridView grid, const int32_t *__restrict__ lowner,
    const int32_t *__restrict__ l2g, int myrank, double tol,
    const double *__restrict__ dest, int32_t *__restrict__ list,
    double *__restrict__ dep, int32_t *__restrict__ eject,
    unsigned long long *__restrict__ ctr,
    unsigned long long *__restrict__ nwalk_ctr) {
  // ctr: [1]=ndep [2]=neject [3]=relocated [4]=loose
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t g = g_lo + blockIdx.x * blockDim.x + threadIdx.x; g < g_hi;
       g += stride) {
    if (!res[g] || !fly[g]) continue;
    if (orig != nullptr && !esc[g]) {
      const Vec3 q{orig[g * 3], orig[g * 3 + 1], orig[g * 3 + 2]};
      const Vec3 p{pos[g * 3], pos[g * 3 + 1], pos[g * 3 + 2]};
      if (q.x != p.x || q.y != p.y || q.z != p.z) {
        atomicAdd(&ctr[3], 1ull);
        bool lo = false;
        const int32_t le = grid_locate(grid, planes, q, tol, &lo);
        if (lo) atomicAdd(&ctr[4], 1ull);
        if (le >= 0) {
          if (lowner[le] == myrank) {
            elem[g] = le;
            pos[g * 3] = q.x;
            pos[g * 3 + 1] = q.y;
            pos[g * 3 + 2] = q.z;
          } else {
            // resampled into a ghost element: reroute to its owner
            const unsigned long long k = atomicAdd(&ctr[1], 1ull);
            dep[k * 9] = (double)g;
            dep[k * 9 + 1] = q.x;
            dep[k * 9 + 2] = q.y;
            dep[k * 9 + 3] = q.z;
            dep[k * 9 + 4] = (double)l2g[le];
            dep[k * 9 + 5] = (double)lowner[le];
            dep[k * 9 + 6] = dest[g * 3];
            dep[k * 9 + 7] = dest[g * 3 + 1];
            dep[k * 9 + 8] = dest[g * 3 + 2];
            res[g] = 0;
            continue;
          }
        } else {
          // not in this rank's submesh at all: host resolves globally
          const unsigned long long k = atomicAdd(&ctr[2], 1ull);
          eject[k] = (int32_t)g;
          continue;
        }
      }
    }
    if (elem[g] < 0) continue; // outside mesh: nothing to walk
    list[g_lo + wave_append(nwalk_ctr)] = (int32_t)g;
  }
}

// use_ovr: round >= 1, every list entry is a fresh arrival whose
// destination came in its record (dest_ovr); round 0 reads the global
// dest by gid.
__global__ void k_part_gather(const int32_t *__restrict__ list, int64_t m,
                              const double *__restrict__ pos,
                              const int32_t *__restrict__ elem,
                              const double *__restrict__ dest,
                              const double *__restrict__ dest_ovr,
                              bool use_ovr,
                              const double *__restrict__ w,
                              const uint16_t *__restrict__ grp,
                              const double *__restrict__ resp, int nscores,
                              double *__restrict__ wpos,
                              double *__restrict__ wdest,
                              int32_t *__restrict__ welem,
                              double *__restrict__ ww,
                              uint16_t *__restrict__ wgrp,
                              double *__restrict__ wresp) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = blockIdx.x * blockDim.x + threadIdx.x; j < m; j += stride) {
    const int64_t g = list[j];
    wpos[j * 3] = pos[g * 3];
    wpos[j * 3 + 1] = pos[g * 3 + 1];
    wpos[j * 3 + 2] = pos[g * 3 + 2];
    const double *dsrc = use_ovr ? dest_ovr : dest;
    wdest[j * 3] = dsrc[g * 3];
    wdest[j * 3 + 1] = dsrc[g * 3 + 1];
    wdest[j * 3 + 2] = dsrc[g * 3 + 2];
    welem[j] = elem[g];
    ww[j] = w[g];
    if (wgrp) wgrp[j] = grp[g];
    if (wresp)
      for (int k = 0; k < nscores; ++k)
        wresp[j * nscores + k] = resp[g * nscores + k];
  }
}

__global__ void k_part_collect(const int32_t *__restrict__ list, int64_t m,
                               const double *__restrict__ wout_pos,
                               const int32_t *__restrict__ wout_elem,
                               const int8_t *__restrict__ wstatus,
                               const double *__restrict__ wout_dest,
                               double *__restrict__ pos,
                               int32_t *__restrict__ elem,
                               uint8_t *__restrict__ esc,
                               uint8_t *__restrict__ res,
                               const int32_t *__restrict__ fgid,
                               const int32_t *__restrict__ fowner,
                               double *__restrict__ dep,
                               unsigned long long *__restrict__ ctr) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = blockIdx.x * blockDim.x + threadIdx.x; j < m; j += stride) {
    const int64_t g = list[j];
    const int8_t st = wstatus[j];
    if (st == 2) {
      const int32_t k = -(wout_elem[j] + 2);
      const unsigned long long d = atomicAdd(&ctr[1], 1ull);
      dep[d * 9] = (double)g;
      dep[d * 9 + 1] = wout_pos[j * 3];
      dep[d * 9 + 2] = wout_pos[j * 3 + 1];
      dep[d * 9 + 3] = wout_pos[j * 3 + 2];
      dep[d * 9 + 4] = (double)fgid[k];
      dep[d * 9 + 5] = (double)fowner[k];
      dep[d * 9 + 6] = wout_dest[j * 3];
      dep[d * 9 + 7] = wout_dest[j * 3 + 1];
      dep[d * 9 + 8] = wout_dest[j * 3 + 2];
      res[g] = 0;
    } else {
      pos[g * 3] = wout_pos[j * 3];
      pos[g * 3 + 1] = wout_pos[j * 3 + 1];
      pos[g * 3 + 2] = wout_pos[j * 3 + 2];
      elem[g] = wout_elem[j];
      esc[g] = (st == 1) ? 1 : 0;
    }
  }
}

__global__ void k_part_count(const double *__restrict__ dep, int64_t m,
                             unsigned long long *__restrict__ dcnt) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride)
    atomicAdd(&dcnt[(int)dep[i * 9 + 5]], 1ull);
}

__global__ void k_part_pack(const double *__restrict__ dep, int64_t m,
                            const int64_t *__restrict__ offs,
                            unsigned long long *__restrict__ cur,
                            double *__restrict__ send) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride) {
    const int o = (int)dep[i * 9 + 5];
    const int64_t s = offs[o] + (int64_t)atomicAdd(&cur[o], 1ull);
    send[s * kRecW] = dep[i * 9];
    send[s * kRecW + 1] = dep[i * 9 + 1];
    send[s * kRecW + 2] = dep[i * 9 + 2];
    send[s * kRecW + 3] = dep[i * 9 + 3];
    send[s * kRecW + 4] = dep[i * 9 + 4];
    send[s * kRecW + 5] = dep[i * 9 + 6];
    send[s * kRecW + 6] = dep[i * 9 + 7];
    send[s * kRecW + 7] = dep[i * 9 + 8];
  }
}

__global__ void k_part_unpack(const double *__restrict__ recv, int64_t m,
                              const int32_t *__restrict__ g2l,
                              double *__restrict__ pos,
                              int32_t *__restrict__ elem,
                              uint8_t *__restrict__ res,
                              uint8_t *__restrict__ esc,
                              double *__restrict__ dest_ovr,
                              int32_t *__restrict__ list,
                              unsigned long long *__restrict__ ctr) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride) {
    const int64_t g = (int64_t)recv[i * kRecW];
    res[g] = 1;
    esc[g] = 0;
    pos[g * 3] = recv[i * kRecW + 1];
    pos[g * 3 + 1] = recv[i * kRecW + 2];
    pos[g * 3 + 2] = recv[i * kRecW + 3];
    elem[g] = g2l[(int64_t)recv[i * kRecW + 4]];
    dest_ovr[g * 3] = recv[i * kRecW + 5];
    dest_ovr[g * 3 + 1] = recv[i * kRecW + 6];
    dest_ovr[g * 3 + 2] = recv[i * kRecW + 7];
    list[wave_append(&ctr[0])] = (int32_t)g;
  }
}

// apply host-resolved out-of-mesh relocations: particle stays resident
// here with elem=-1 at its requested origin
__global__ void k_part_apply_outside(const int32_t *__restrict__ gids,
                                     int64_t m,
                                     const double *__restrict__ orig,
                                     double *__restrict__ pos,
                                     int32_t *__restrict__ elem) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride) {
    const int64_t g = gids[i];
    elem[g] = -1;
    pos[g * 3] = orig[g * 3];
    pos[g * 3 + 1] = orig[g * 3 + 1];
    pos[g * 3 + 2] = orig[g * 3 + 2];
  }
}

template <class T> T *pdmalloc(int64_t count) {
  void *p = nullptr;
  PT_HIP_CHECK(hipMalloc(&p, count * sizeof(T)));
  return (T *)p;
}

// ---------------------------------------------------------------------------
// shared host-side decomposition setup
// ---------------------------------------------------------------------------

struct Decomp {
  SubMesh sub;
  std::vector<int32_t> owners;      // global elem -> rank
  std::vector<int32_t> l2g32;       // local elem -> global (int32)
  std::vector<int32_t> lowner;      // local elem -> owner rank
  std::vector<int32_t> g2l;         // global elem -> local (-1 if absent)
  std::vector<int64_t> l2g() const { return sub.elem_l2g; }
};

Decomp build_decomp(const Mesh &full, int rank, int world,
                    const int32_t *owners_in, int ghost_rings) {
  if (full.has_periodic() && world > 1)
    throw std::runtime_error(
        "PartitionedEngine: periodic BCs with world > 1 are unsupported in "
        "the stateful engine (a cross-rank wrap would need a translated "
        "destination, but destinations are gathered by gid from the "
        "untranslated global arrays); use the stateless "
        "pumiumtally_amd.parallel.PartitionedTally driver, which ships the "
        "translated destination in its exchange records");
  Decomp d;
  d.owners = owners_in
                 ? std::vector<int32_t>(owners_in, owners_in + full.nelems)
                 : partition_morton(full, world);
  d.sub = extract_submesh(full, d.owners, rank, ghost_rings);
  const int64_t nl = d.sub.local.nelems;
  d.l2g32.resize(nl);
  d.lowner.resize(nl);
  for (int64_t t = 0; t < nl; ++t) {
    d.l2g32[t] = (int32_t)d.sub.elem_l2g[t];
    d.lowner[t] = d.owners[d.sub.elem_l2g[t]];
  }
  d.g2l.assign(full.nelems, -1);
  for (int64_t t = 0; t < nl; ++t) d.g2l[d.sub.elem_l2g[t]] = (int32_t)t;
  return d;
}

// ---------------------------------------------------------------------------
// GPU implementation
// ---------------------------------------------------------------------------

class GpuPartitionedEngine final : public PartitionedEngine {
public:
  GpuPartitionedEngine(const Mesh &full, int64_t n, Comm *comm, int rank,
                       int world, int device, int ngroups, int nscores,
                       const int32_t *owners, int ghost_rings)
      : n_(n), comm_(comm), rank_(rank), world_(world), device_(device),
        ngroups_(ngroups < 1 ? 1 : ngroups),
        nscores_(nscores < 1 ? 1 : nscores), nelems_global_(full.nelems),
        dec_(build_decomp(full, rank, world, owners, ghost_rings)) {
    if (world_ > 1 && !comm_)
      throw std::runtime_error("PartitionedEngine: world > 1 needs a comm");
    PT_HIP_CHECK(hipSetDevice(device_));
    eng_ = make_gpu_engine(dec_.sub.local, 1, device_, ngroups_, nscores_);
    if (!eng_) throw std::runtime_error("no HIP device for PartitionedEngine");
    if (!eng_->device_mesh(&dmesh_))
      throw std::runtime_error("GPU engine did not expose its device mesh");
    cs_ = (hipStream_t)dmesh_.stream; // the engine's compute stream
    PT_HIP_CHECK(hipStreamCreateWithFlags(&s_copy_, hipStreamNonBlocking));
    for (auto &ev : ev_in_)
      PT_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    loc_tol_ = loc_tol_rel() * norm(full.bbox_hi - full.bbox_lo);

    // lookup tables
    const int64_t nl = dec_.sub.local.nelems;
    d_lowner_ = pdmalloc<int32_t>(nl);
    d_l2g_ = pdmalloc<int32_t>(nl);
    PT_HIP_CHECK(hipMemcpy(d_lowner_, dec_.lowner.data(), nl * 4,
                           hipMemcpyHostToDevice));
    PT_HIP_CHECK(
        hipMemcpy(d_l2g_, dec_.l2g32.data(), nl * 4, hipMemcpyHostToDevice));
    d_g2l_ = pdmalloc<int32_t>(nelems_global_);
    PT_HIP_CHECK(hipMemcpy(d_g2l_, dec_.g2l.data(), nelems_global_ * 4,
                           hipMemcpyHostToDevice));
    if (!dec_.sub.foreign_gid.empty()) {
      const int64_t nf = (int64_t)dec_.sub.foreign_gid.size();
      std::vector<int32_t> fg(nf);
      for (int64_t k = 0; k < nf; ++k) fg[k] = (int32_t)dec_.sub.foreign_gid[k];
      d_fgid_ = pdmalloc<int32_t>(nf);
      d_fowner_ = pdmalloc<int32_t>(nf);
      PT_HIP_CHECK(hipMemcpy(d_fgid_, fg.data(), nf * 4,
                             hipMemcpyHostToDevice));
      PT_HIP_CHECK(hipMemcpy(d_fowner_, dec_.sub.foreign_owner.data(), nf * 4,
                             hipMemcpyHostToDevice));
    }

    // per-global-particle state
    d_pos_ = pdmalloc<double>(n_ * 3);
    d_elem_ = pdmalloc<int32_t>(n_);
    d_res_ = pdmalloc<uint8_t>(n_);
    d_esc_ = pdmalloc<uint8_t>(n_);
    PT_HIP_CHECK(hipMemset(d_res_, 0, n_));
    PT_HIP_CHECK(hipMemset(d_esc_, 0, n_));

    // step inputs (global)
    d_dest_ = pdmalloc<double>(n_ * 3);
    d_fly_ = pdmalloc<int8_t>(n_);
    d_w_ = pdmalloc<double>(n_);

    // work buffers
    d_list_ = pdmalloc<int32_t>(n_);
    d_dep_ = pdmalloc<double>(n_ * 9);
    d_dest_ovr_ = pdmalloc<double>(n_ * 3);
    d_eject_ = pdmalloc<int32_t>(n_);
    // counters: [0..4] round/step counters, [8 .. 8+2*world) bucket
    // counts+cursors, then kMaxChunks per-chunk walk counters
    ctr_chunk0_ = 8 + 2 * world_;
    d_ctr_ = pdmalloc<unsigned long long>(ctr_chunk0_ + kMaxChunks);
    d_wpos_ = pdmalloc<double>(n_ * 3);
    d_wdest_ = pdmalloc<double>(n_ * 3);
    d_welem_ = pdmalloc<int32_t>(n_);
    d_ww_ = pdmalloc<double>(n_);
    d_wout_pos_ = pdmalloc<double>(n_ * 3);
    d_wout_dest_ = pdmalloc<double>(n_ * 3);
    d_wout_elem_ = pdmalloc<int32_t>(n_);
    d_wstatus_ = pdmalloc<int8_t>(n_);
    d_offs_ = pdmalloc<int64_t>(world_);
  }

  ~GpuPartitionedEngine() override {
    (void)hipSetDevice(device_);
    (void)hipDeviceSynchronize();
    for (auto &ev : ev_in_) (void)hipEventDestroy(ev);
    (void)hipStreamDestroy(s_copy_);
    for (void *p :
         {(void *)d_lowner_, (void *)d_l2g_, (void *)d_g2l_, (void *)d_fgid_,
          (void *)d_fowner_, (void *)d_pos_, (void *)d_elem_, (void *)d_res_,
          (void *)d_esc_, (void *)d_dest_, (void *)d_fly_, (void *)d_w_,
          (void *)d_grp_, (void *)d_orig_, (void *)d_list_, (void *)d_dep_, (void *)d_dest_ovr_,
          (void *)d_eject_, (void *)d_ctr_, (void *)d_wpos_, (void *)d_wdest_,
          (void *)d_welem_, (void *)d_ww_, (void *)d_wgrp_,
          (void *)d_wout_pos_, (void *)d_wout_dest_,
          (void *)d_wout_elem_, (void *)d_wstatus_,
          (void *)d_offs_, (void *)d_send_, (void *)d_recv_,
          (void *)d_resp_, (void *)d_wresp_})
      if (p) (void)hipFree(p);
  }

  int rank() const override { return rank_; }
  int world() const override { return world_; }
  int64_t num_particles() const override { return n_; }

  void localize(const double *origins, int64_t n) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    eng_->synchronize();
    PT_HIP_CHECK(
        hipMemcpy(d_dest_, origins, n_ * 3 * 8, hipMemcpyHostToDevice));
    const int64_t nwords = (n_ + 63) / 64;
    unsigned long long *d_claim = pdmalloc<unsigned long long>(nwords);
    PT_HIP_CHECK(hipMemset(d_claim, 0, nwords * 8));
    PT_HIP_CHECK(hipMemset(d_ctr_, 0, 8 * 8));
    k_part_localize<<<pgrid(n_), kPBlock>>>(
        dmesh_.planes, dmesh_.grid, d_lowner_, rank_, d_dest_, n_, loc_tol_,
        d_pos_, d_elem_, d_res_, d_esc_, d_claim, &d_ctr_[4]);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipDeviceSynchronize());
    if (world_ > 1) {
      // claims are disjoint (only the owner claims), so bitwise OR of the
      // claim masks == integer sum of the words
      std::vector<int64_t> words(nwords);
      PT_HIP_CHECK(hipMemcpy(words.data(), d_claim, nwords * 8,
                             hipMemcpyDeviceToHost));
      comm_->allreduce_sum(words.data(), nwords);
      PT_HIP_CHECK(hipMemcpy(d_claim, words.data(), nwords * 8,
                             hipMemcpyHostToDevice));
    }
    if (rank_ == 0) {
      k_part_claim_rest<<<pgrid(n_), kPBlock>>>(d_claim, d_dest_, n_, d_pos_,
                                                d_elem_, d_res_, d_esc_);
      PT_HIP_CHECK(hipGetLastError());
    }
    PT_HIP_CHECK(hipDeviceSynchronize());
    PT_HIP_CHECK(hipFree(d_claim));
    unsigned long long loose = 0;
    PT_HIP_CHECK(hipMemcpy(&loose, &d_ctr_[4], 8, hipMemcpyDeviceToHost));
    stats_.loose_localizations += (int64_t)loose;
  }

  void step(const double *dest, const int8_t *flying, const double *weights,
            int64_t n, const double *origin, const uint16_t *groups,
            const double *responses) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    if (groups && ngroups_ <= 1)
      throw std::runtime_error("groups passed but ngroups == 1");
    if (responses && !d_resp_) {
      d_resp_ = pdmalloc<double>(n_ * nscores_);
      d_wresp_ = pdmalloc<double>(n_ * nscores_);
    }
    eng_->synchronize();
    if (origin && !d_orig_) d_orig_ = pdmalloc<double>(n_ * 3);
    if (groups && !d_grp_) {
      d_grp_ = pdmalloc<uint16_t>(n_);
      d_wgrp_ = pdmalloc<uint16_t>(n_);
    }

    // Chunked copy/walk pipeline: all chunks' H2D copies are enqueued up
    // front on the copy stream; chunk c's prepare/gather/walk on the
    // compute stream waits only for chunk c's event, so the walk of
    // chunk c overlaps the copies of chunks c+1..  (measured: the
    // monolithic step serialized ~5.9 ms of H2D against ~6.5 ms of
    // kernels).
    const int C = (int)std::min<int64_t>(
        kMaxChunks, std::max<int64_t>(1, n_ >> 20)); // >=1M particles/chunk
    PT_HIP_CHECK(hipMemsetAsync(d_ctr_, 0,
                                (ctr_chunk0_ + kMaxChunks) * 8, cs_));
    std::vector<int64_t> clo(C + 1);
    for (int c = 0; c <= C; ++c) clo[c] = n_ * c / C;
    for (int c = 0; c < C; ++c) {
      const int64_t lo = clo[c], hi = clo[c + 1];
      PT_HIP_CHECK(hipMemcpyAsync(d_fly_ + lo, flying + lo, hi - lo,
                                  hipMemcpyHostToDevice, s_copy_));
      PT_HIP_CHECK(hipMemcpyAsync(d_w_ + lo, weights + lo, (hi - lo) * 8,
                                  hipMemcpyHostToDevice, s_copy_));
      PT_HIP_CHECK(hipMemcpyAsync(d_dest_ + lo * 3, dest + lo * 3,
                                  (hi - lo) * 24, hipMemcpyHostToDevice,
                                  s_copy_));
      if (origin)
        PT_HIP_CHECK(hipMemcpyAsync(d_orig_ + lo * 3, origin + lo * 3,
                                    (hi - lo) * 24, hipMemcpyHostToDevice,
                                    s_copy_));
      if (groups)
        PT_HIP_CHECK(hipMemcpyAsync(d_grp_ + lo, groups + lo, (hi - lo) * 2,
                                    hipMemcpyHostToDevice, s_copy_));
      if (responses)
        PT_HIP_CHECK(hipMemcpyAsync(d_resp_ + lo * nscores_,
                                    responses + lo * nscores_,
                                    (hi - lo) * nscores_ * 8,
                                    hipMemcpyHostToDevice, s_copy_));
      PT_HIP_CHECK(hipEventRecord(ev_in_[c], s_copy_));
    }

    int64_t nwalk = 0; // round-0 total across chunks (for stats only)
    for (int c = 0; c < C; ++c) {
      const int64_t lo = clo[c], hi = clo[c + 1];
      PT_HIP_CHECK(hipStreamWaitEvent(cs_, ev_in_[c], 0));
      k_part_prepare<<<pgrid(hi - lo), kPBlock, 0, cs_>>>(
          lo, hi, d_res_, d_esc_, d_fly_, origin ? d_orig_ : nullptr,
          d_pos_, d_elem_, dmesh_.planes, dmesh_.grid, d_lowner_, d_l2g_,
          rank_, loc_tol_, d_dest_, d_list_, d_dep_, d_eject_, d_ctr_,
          &d_ctr_[ctr_chunk0_ + c]);
      PT_HIP_CHECK(hipGetLastError());
      PT_HIP_CHECK(hipStreamSynchronize(cs_));
      unsigned long long cw = 0;
      PT_HIP_CHECK(hipMemcpy(&cw, &d_ctr_[ctr_chunk0_ + c], 8,
                             hipMemcpyDeviceToHost));
      if (cw == 0) continue;
      nwalk += (int64_t)cw;
      k_part_gather<<<pgrid((int64_t)cw), kPBlock, 0, cs_>>>(
          d_list_ + lo, (int64_t)cw, d_pos_, d_elem_, d_dest_, d_dest_ovr_,
          /*use_ovr=*/false, d_w_, d_grp_, d_resp_, nscores_, d_wpos_,
          d_wdest_, d_welem_, d_ww_, groups ? d_wgrp_ : nullptr,
          responses ? d_wresp_ : nullptr);
      PT_HIP_CHECK(hipGetLastError());
      eng_->walk_raw_device((int64_t)cw, d_wpos_, d_wdest_, d_welem_,
                            d_ww_, d_wout_pos_, d_wout_elem_, d_wstatus_,
                            groups ? d_wgrp_ : nullptr,
                            responses ? d_wresp_ : nullptr, d_wout_dest_);
      k_part_collect<<<pgrid((int64_t)cw), kPBlock, 0, cs_>>>(
          d_list_ + lo, (int64_t)cw, d_wout_pos_, d_wout_elem_, d_wstatus_,
          d_wout_dest_, d_pos_, d_elem_, d_esc_, d_res_, d_fgid_,
          d_fowner_, d_dep_, d_ctr_);
      PT_HIP_CHECK(hipGetLastError());
    }
    PT_HIP_CHECK(hipStreamSynchronize(cs_));
    (void)nwalk;

    unsigned long long hctr[5];
    PT_HIP_CHECK(hipMemcpy(hctr, d_ctr_, 5 * 8, hipMemcpyDeviceToHost));
    // host-resolve ejected relocations (rare: resampled origins that
    // missed the local submesh grid); their reroute records join the dep
    // list before the first exchange below
    if (hctr[2] > 0) host_resolve_ejects((int64_t)hctr[2], origin, dest);
    stats_.relocated += (int64_t)hctr[3];
    stats_.loose_localizations += (int64_t)hctr[4];

    // rounds 1+: walk lists come from exchanged records (round 0's walks
    // already ran chunk-pipelined above)
    int64_t nwalk_r = 0;
    for (int round = 1; round <= max_rounds_; ++round) {
      if (nwalk_r > 0) {
        k_part_gather<<<pgrid(nwalk_r), kPBlock, 0, cs_>>>(
            d_list_, nwalk_r, d_pos_, d_elem_, d_dest_, d_dest_ovr_,
            /*use_ovr=*/true, d_w_, d_grp_, d_resp_,
            nscores_, d_wpos_, d_wdest_, d_welem_, d_ww_,
            groups ? d_wgrp_ : nullptr, responses ? d_wresp_ : nullptr);
        PT_HIP_CHECK(hipGetLastError());
        eng_->walk_raw_device(nwalk_r, d_wpos_, d_wdest_, d_welem_, d_ww_,
                              d_wout_pos_, d_wout_elem_, d_wstatus_,
                              groups ? d_wgrp_ : nullptr,
                              responses ? d_wresp_ : nullptr, d_wout_dest_);
        k_part_collect<<<pgrid(nwalk_r), kPBlock, 0, cs_>>>(
            d_list_, nwalk_r, d_wout_pos_, d_wout_elem_, d_wstatus_,
            d_wout_dest_, d_pos_, d_elem_, d_esc_, d_res_, d_fgid_,
            d_fowner_, d_dep_, d_ctr_);
        PT_HIP_CHECK(hipGetLastError());
        PT_HIP_CHECK(hipStreamSynchronize(cs_));
      }
      unsigned long long ndep = 0;
      PT_HIP_CHECK(
          hipMemcpy(&ndep, &d_ctr_[1], 8, hipMemcpyDeviceToHost));
      const int64_t m = (int64_t)ndep;

      // per-destination counts -> offsets -> packed send buffer
      std::vector<int64_t> scounts(world_, 0);
      if (m > 0) {
        PT_HIP_CHECK(hipMemsetAsync(&d_ctr_[8], 0, world_ * 8, cs_));
        k_part_count<<<pgrid(m), kPBlock, 0, cs_>>>(d_dep_, m, &d_ctr_[8]);
        PT_HIP_CHECK(hipGetLastError());
        PT_HIP_CHECK(hipStreamSynchronize(cs_));
        std::vector<unsigned long long> dc(world_);
        PT_HIP_CHECK(hipMemcpy(dc.data(), &d_ctr_[8], world_ * 8,
                               hipMemcpyDeviceToHost));
        std::vector<int64_t> offs(world_);
        int64_t acc = 0;
        for (int r = 0; r < world_; ++r) {
          offs[r] = acc;
          scounts[r] = (int64_t)dc[r];
          acc += (int64_t)dc[r];
        }
        ensure_cap(&d_send_, &cap_send_, m * kRecW);
        PT_HIP_CHECK(hipMemcpy(d_offs_, offs.data(), world_ * 8,
                               hipMemcpyHostToDevice));
        PT_HIP_CHECK(hipMemsetAsync(&d_ctr_[8 + world_], 0, world_ * 8,
                                    cs_));
        k_part_pack<<<pgrid(m), kPBlock, 0, cs_>>>(d_dep_, m, d_offs_,
                                                   &d_ctr_[8 + world_],
                                                   d_send_);
        PT_HIP_CHECK(hipGetLastError());
        PT_HIP_CHECK(hipStreamSynchronize(cs_));
      }

      // global termination + recv counts
      int64_t nrecv = 0;
      const double *recv_ptr = nullptr;
      if (world_ == 1) {
        if (m == 0) break;
        recv_ptr = d_send_; // self-exchange
        nrecv = m;
      } else {
        std::vector<int64_t> flat((int64_t)world_ * world_, 0);
        for (int r = 0; r < world_; ++r)
          flat[(int64_t)rank_ * world_ + r] = scounts[r];
        comm_->allreduce_sum(flat.data(), (int64_t)world_ * world_);
        int64_t total = 0;
        for (int64_t c : flat) total += c;
        if (total == 0) break;
        std::vector<int64_t> rcounts(world_);
        for (int s = 0; s < world_; ++s)
          rcounts[s] = flat[(int64_t)s * world_ + rank_];
        // counts are in records; collectives speak doubles
        std::vector<int64_t> sc(world_), rc(world_);
        for (int r = 0; r < world_; ++r) {
          sc[r] = scounts[r] * kRecW;
          rc[r] = rcounts[r] * kRecW;
        }
        for (int64_t c : rcounts) nrecv += c;
        if (comm_->has_device_collectives()) {
          double *d_recv = nullptr;
          comm_->alltoallv_device(d_send_, sc, rc, &d_recv);
          recv_ptr = d_recv;
        } else {
          // host-staged exchange (TCP fallback): D2H -> alltoallv -> H2D
          std::vector<double> hsend(m * kRecW);
          if (m)
            PT_HIP_CHECK(hipMemcpy(hsend.data(), d_send_, m * kRecW * 8,
                                   hipMemcpyDeviceToHost));
          std::vector<double> hrecv = comm_->alltoallv(hsend.data(), sc);
          ensure_cap(&d_recv_, &cap_recv_, (int64_t)hrecv.size());
          if (!hrecv.empty())
            PT_HIP_CHECK(hipMemcpy(d_recv_, hrecv.data(), hrecv.size() * 8,
                                   hipMemcpyHostToDevice));
          recv_ptr = d_recv_;
        }
      }

      // unpack received records; they form the next round's walk list
      PT_HIP_CHECK(hipMemsetAsync(d_ctr_, 0, 2 * 8, cs_)); // nwalk, ndep
      if (nrecv > 0) {
        k_part_unpack<<<pgrid(nrecv), kPBlock, 0, cs_>>>(
            recv_ptr, nrecv, d_g2l_, d_pos_, d_elem_, d_res_, d_esc_,
            d_dest_ovr_, d_list_, d_ctr_);
        PT_HIP_CHECK(hipGetLastError());
      }
      PT_HIP_CHECK(hipStreamSynchronize(cs_));
      nwalk_r = nrecv;
      if (round == max_rounds_)
        throw std::runtime_error("partitioned step did not converge in " +
                                 std::to_string(max_rounds_) +
                                 " handoff rounds");
    }
    stats_.moves++;
  }

  std::vector<double> flux_global() override {
    eng_->synchronize();
    const std::vector<double> local = eng_->flux(); // nscores*ngroups*nlocal
    const int64_t nl = dec_.sub.local.nelems;
    const int64_t slabs = (int64_t)nscores_ * ngroups_;
    std::vector<double> out(slabs * nelems_global_, 0.0);
    for (int64_t s = 0; s < slabs; ++s)
      for (int64_t t = 0; t < nl; ++t)
        out[s * nelems_global_ + dec_.sub.elem_l2g[t]] +=
            local[s * nl + t];
    if (world_ > 1)
      comm_->allreduce_sum(out.data(), (int64_t)out.size());
    return out;
  }

  int64_t resident() const override {
    PT_HIP_CHECK(hipSetDevice(device_));
    eng_->synchronize(); // order the NULL-stream copy behind cs_ work
    std::vector<uint8_t> r(n_);
    PT_HIP_CHECK(hipMemcpy(r.data(), d_res_, n_, hipMemcpyDeviceToHost));
    int64_t c = 0;
    for (uint8_t v : r) c += v;
    return c;
  }

  const EngineStats &stats() const override {
    const EngineStats &inner = eng_->stats();
    stats_.lost_particles = inner.lost_particles;
    return stats_;
  }

  void synchronize() override { eng_->synchronize(); }

  std::vector<uint8_t> resident_mask() const override {
    eng_->synchronize();
    std::vector<uint8_t> r(n_);
    PT_HIP_CHECK(hipMemcpy(r.data(), d_res_, n_, hipMemcpyDeviceToHost));
    return r;
  }
  std::vector<double> positions() const override {
    eng_->synchronize();
    std::vector<double> p(n_ * 3);
    PT_HIP_CHECK(hipMemcpy(p.data(), d_pos_, n_ * 24, hipMemcpyDeviceToHost));
    return p;
  }
  std::vector<int32_t> elem_ids() const override {
    eng_->synchronize();
    std::vector<int32_t> e(n_);
    PT_HIP_CHECK(hipMemcpy(e.data(), d_elem_, n_ * 4, hipMemcpyDeviceToHost));
    return e;
  }

private:
  void check_n(int64_t n) const {
    if (n != n_) throw std::runtime_error("global particle count mismatch");
  }

  void ensure_cap(double **p, int64_t *cap, int64_t n) {
    if (n <= *cap) return;
    if (*p) PT_HIP_CHECK(hipFree(*p));
    *cap = n + n / 4;
    PT_HIP_CHECK(hipMalloc((void **)p, *cap * 8));
  }

  // Rare path: a resampled origin missed the local submesh grid.  The
  // full mesh stays on the host (the caller owns it; we keep a copy of
  // the pieces needed): resolve globally, then inject reroute records
  // into the departure list (self-routes included -- they come back
  // through the exchange uniformly).
  void host_resolve_ejects(int64_t ne, const double *origin,
                           const double *dest_host) {
    std::vector<int32_t> gids(ne);
    PT_HIP_CHECK(
        hipMemcpy(gids.data(), d_eject_, ne * 4, hipMemcpyDeviceToHost));
    std::vector<double> dep9;
    std::vector<int32_t> outside;
    for (int64_t i = 0; i < ne; ++i) {
      const int64_t g = gids[i];
      const Vec3 q{origin[g * 3], origin[g * 3 + 1], origin[g * 3 + 2]};
      bool lo = false;
      const int32_t ge = full_locate_ ? full_locate_(q, loc_tol_, &lo) : -1;
      if (lo) stats_.loose_localizations++;
      if (ge >= 0) {
        dep9.insert(dep9.end(),
                    {(double)g, q.x, q.y, q.z, (double)ge,
                     (double)dec_.owners[ge], dest_host[g * 3],
                     dest_host[g * 3 + 1], dest_host[g * 3 + 2]});
      } else {
        outside.push_back((int32_t)g);
      }
    }
    if (!dep9.empty()) {
      // append to the device dep list (capacity n_*9 is plenty: ejects
      // are a subset of residents)
      unsigned long long ndep = 0;
      PT_HIP_CHECK(hipMemcpy(&ndep, &d_ctr_[1], 8, hipMemcpyDeviceToHost));
      const int64_t m = (int64_t)dep9.size() / 9;
      PT_HIP_CHECK(hipMemcpy(d_dep_ + (int64_t)ndep * 9, dep9.data(),
                             dep9.size() * 8, hipMemcpyHostToDevice));
      ndep += (unsigned long long)m;
      PT_HIP_CHECK(hipMemcpy(&d_ctr_[1], &ndep, 8, hipMemcpyHostToDevice));
      // the particle left this rank
      std::vector<uint8_t> zero(1, 0);
      for (int64_t i = 0; i < m; ++i) {
        const int64_t g = (int64_t)dep9[i * 9];
        PT_HIP_CHECK(
            hipMemcpy(d_res_ + g, zero.data(), 1, hipMemcpyHostToDevice));
      }
    }
    if (!outside.empty()) {
      int32_t *d_o = pdmalloc<int32_t>((int64_t)outside.size());
      PT_HIP_CHECK(hipMemcpy(d_o, outside.data(), outside.size() * 4,
                             hipMemcpyHostToDevice));
      k_part_apply_outside<<<pgrid((int64_t)outside.size()), kPBlock>>>(
          d_o, (int64_t)outside.size(), d_orig_, d_pos_, d_elem_);
      PT_HIP_CHECK(hipGetLastError());
      PT_HIP_CHECK(hipDeviceSynchronize());
      PT_HIP_CHECK(hipFree(d_o));
    }
  }

public:
  // bound by the factory: global locate over the full mesh (host)
  std::function<int32_t(Vec3, double, bool *)> full_locate_;
  int max_rounds_ = 64;

private:
  int64_t n_;
  Comm *comm_;
  int rank_, world_, device_;
  int ngroups_, nscores_ = 1;
  int64_t nelems_global_;
  Decomp dec_;
  std::unique_ptr<Engine> eng_;
  Engine::DeviceMeshView dmesh_{};
  hipStream_t cs_ = nullptr;     // the inner engine's compute stream
  hipStream_t s_copy_ = nullptr; // step-input H2D pipeline
  hipEvent_t ev_in_[kMaxChunks] = {};
  int ctr_chunk0_ = 8;
  double loc_tol_ = 0.0;
  mutable EngineStats stats_;

  int32_t *d_lowner_ = nullptr, *d_l2g_ = nullptr, *d_g2l_ = nullptr;
  int32_t *d_fgid_ = nullptr, *d_fowner_ = nullptr;
  double *d_pos_ = nullptr;
  int32_t *d_elem_ = nullptr;
  uint8_t *d_res_ = nullptr, *d_esc_ = nullptr;
  double *d_dest_ = nullptr, *d_w_ = nullptr, *d_orig_ = nullptr;
  double *d_resp_ = nullptr, *d_wresp_ = nullptr;
  int8_t *d_fly_ = nullptr;
  uint16_t *d_grp_ = nullptr, *d_wgrp_ = nullptr;
  int32_t *d_list_ = nullptr, *d_eject_ = nullptr;
  double *d_dep_ = nullptr;
  unsigned long long *d_ctr_ = nullptr;
  double *d_wpos_ = nullptr, *d_wdest_ = nullptr, *d_ww_ = nullptr;
  int32_t *d_welem_ = nullptr, *d_wout_elem_ = nullptr;
  double *d_wout_pos_ = nullptr, *d_wout_dest_ = nullptr;
  double *d_dest_ovr_ = nullptr;
  int8_t *d_wstatus_ = nullptr;
  int64_t *d_offs_ = nullptr;
  double *d_send_ = nullptr, *d_recv_ = nullptr;
  int64_t cap_send_ = 0, cap_recv_ = 0;
};

// ---------------------------------------------------------------------------
// CPU implementation (oracle / fallback): same algorithm, host loops
// ---------------------------------------------------------------------------

class CpuPartitionedEngine final : public PartitionedEngine {
public:
  CpuPartitionedEngine(const Mesh &full, int64_t n, Comm *comm, int rank,
                       int world, int ngroups, int nscores,
                       const int32_t *owners, int ghost_rings)
      : n_(n), comm_(comm), rank_(rank), world_(world),
        ngroups_(ngroups < 1 ? 1 : ngroups),
        nscores_(nscores < 1 ? 1 : nscores), nelems_global_(full.nelems),
        dec_(build_decomp(full, rank, world, owners, ghost_rings)) {
    if (world_ > 1 && !comm_)
      throw std::runtime_error("PartitionedEngine: world > 1 needs a comm");
    eng_ = make_cpu_engine(dec_.sub.local, 1, ngroups_, nscores_);
    loc_tol_ = loc_tol_rel() * norm(full.bbox_hi - full.bbox_lo);
    pos_.assign(n_ * 3, 0.0);
    elem_.assign(n_, -1);
    res_.assign(n_, 0);
    esc_.assign(n_, 0);
  }

  int rank() const override { return rank_; }
  int world() const override { return world_; }
  int64_t num_particles() const override { return n_; }

  void localize(const double *origins, int64_t n) override {
    check_n(n);
    const Mesh &lm = dec_.sub.local;
    std::vector<int64_t> claim((n_ + 63) / 64, 0);
    for (int64_t g = 0; g < n_; ++g) {
      res_[g] = 0;
      const Vec3 q{origins[g * 3], origins[g * 3 + 1], origins[g * 3 + 2]};
      bool lo = false;
      const int32_t le = lm.locate(q, loc_tol_, &lo);
      if (le >= 0 && dec_.lowner[le] == rank_) {
        if (lo) stats_.loose_localizations++;
        res_[g] = 1;
        esc_[g] = 0;
        elem_[g] = le;
        pos_[g * 3] = q.x;
        pos_[g * 3 + 1] = q.y;
        pos_[g * 3 + 2] = q.z;
        claim[g >> 6] |= (int64_t)(1ull << (g & 63));
      }
    }
    if (world_ > 1)
      comm_->allreduce_sum(claim.data(), (int64_t)claim.size());
    if (rank_ == 0) {
      for (int64_t g = 0; g < n_; ++g)
        if (!((claim[g >> 6] >> (g & 63)) & 1)) {
          res_[g] = 1;
          esc_[g] = 0;
          elem_[g] = -1;
          pos_[g * 3] = origins[g * 3];
          pos_[g * 3 + 1] = origins[g * 3 + 1];
          pos_[g * 3 + 2] = origins[g * 3 + 2];
        }
    }
  }

  void step(const double *dest, const int8_t *flying, const double *weights,
            int64_t n, const double *origin, const uint16_t *groups,
            const double *responses) override {
    check_n(n);
    if (groups && ngroups_ <= 1)
      throw std::runtime_error("groups passed but ngroups == 1");
    const Mesh &lm = dec_.sub.local;
    std::vector<int32_t> list;
    std::vector<double> dep; // 9 doubles per entry (see kRecW comment)
    std::vector<double> ovr; // per-arrival destination overrides
    for (int64_t g = 0; g < n_; ++g) {
      if (!res_[g] || !flying[g]) continue;
      if (origin && !esc_[g]) {
        const Vec3 q{origin[g * 3], origin[g * 3 + 1], origin[g * 3 + 2]};
        if (q.x != pos_[g * 3] || q.y != pos_[g * 3 + 1] ||
            q.z != pos_[g * 3 + 2]) {
          stats_.relocated++;
          bool lo = false;
          int32_t le = lm.locate(q, loc_tol_, &lo);
          if (lo) stats_.loose_localizations++;
          if (le >= 0 && dec_.lowner[le] != rank_) {
            dep.insert(dep.end(), {(double)g, q.x, q.y, q.z,
                                   (double)dec_.l2g32[le],
                                   (double)dec_.lowner[le], dest[g * 3],
                                   dest[g * 3 + 1], dest[g * 3 + 2]});
            res_[g] = 0;
            continue;
          }
          if (le < 0) {
            // global resolve via full-mesh locate
            bool lo2 = false;
            const int32_t ge =
                full_locate_ ? full_locate_(q, loc_tol_, &lo2) : -1;
            if (lo2) stats_.loose_localizations++;
            if (ge >= 0) {
              dep.insert(dep.end(), {(double)g, q.x, q.y, q.z, (double)ge,
                                     (double)dec_.owners[ge], dest[g * 3],
                                     dest[g * 3 + 1], dest[g * 3 + 2]});
              res_[g] = 0;
            } else {
              elem_[g] = -1;
              pos_[g * 3] = q.x;
              pos_[g * 3 + 1] = q.y;
              pos_[g * 3 + 2] = q.z;
            }
            continue;
          }
          elem_[g] = le;
          pos_[g * 3] = q.x;
          pos_[g * 3 + 1] = q.y;
          pos_[g * 3 + 2] = q.z;
        }
      }
      if (elem_[g] < 0) continue;
      list.push_back((int32_t)g);
    }

    for (int round = 0; round < max_rounds_; ++round) {
      if (!list.empty()) {
        const int64_t m = (int64_t)list.size();
        std::vector<double> wpos(m * 3), wdest(m * 3), ww(m), wout_pos(m * 3);
        std::vector<double> wout_dest(m * 3);
        std::vector<int32_t> welem(m), wout_elem(m);
        std::vector<int8_t> wstatus(m);
        std::vector<uint16_t> wgrp(groups ? m : 0);
        std::vector<double> wresp(responses ? m * nscores_ : 0);
        const double *dsrc = round > 0 ? ovr.data() : dest;
        for (int64_t j = 0; j < m; ++j) {
          const int64_t g = list[j];
          for (int k = 0; k < 3; ++k) {
            wpos[j * 3 + k] = pos_[g * 3 + k];
            wdest[j * 3 + k] = dsrc[g * 3 + k];
          }
          welem[j] = elem_[g];
          ww[j] = weights[g];
          if (groups) wgrp[j] = groups[g];
          if (responses)
            for (int k = 0; k < nscores_; ++k)
              wresp[j * nscores_ + k] = responses[g * nscores_ + k];
        }
        eng_->walk_raw(m, wpos.data(), wdest.data(), welem.data(), ww.data(),
                       wout_pos.data(), wout_elem.data(), wstatus.data(),
                       groups ? wgrp.data() : nullptr,
                       responses ? wresp.data() : nullptr, wout_dest.data());
        for (int64_t j = 0; j < m; ++j) {
          const int64_t g = list[j];
          if (wstatus[j] == 2) {
            const int32_t k = -(wout_elem[j] + 2);
            dep.insert(dep.end(),
                       {(double)g, wout_pos[j * 3], wout_pos[j * 3 + 1],
                        wout_pos[j * 3 + 2],
                        (double)dec_.sub.foreign_gid[k],
                        (double)dec_.sub.foreign_owner[k],
                        wout_dest[j * 3], wout_dest[j * 3 + 1],
                        wout_dest[j * 3 + 2]});
            res_[g] = 0;
          } else {
            for (int k = 0; k < 3; ++k) pos_[g * 3 + k] = wout_pos[j * 3 + k];
            elem_[g] = wout_elem[j];
            esc_[g] = (wstatus[j] == 1) ? 1 : 0;
          }
        }
      }
      list.clear();

      // bucket by destination, exchange, unpack
      const int64_t m = (int64_t)dep.size() / 9;
      std::vector<int64_t> scounts(world_, 0);
      std::vector<double> send(m * kRecW);
      {
        std::vector<int64_t> offs(world_, 0), cur(world_, 0);
        for (int64_t i = 0; i < m; ++i) scounts[(int)dep[i * 9 + 5]]++;
        int64_t acc = 0;
        for (int r = 0; r < world_; ++r) {
          offs[r] = acc;
          acc += scounts[r];
        }
        for (int64_t i = 0; i < m; ++i) {
          const int o = (int)dep[i * 9 + 5];
          const int64_t s = offs[o] + cur[o]++;
          for (int k = 0; k < 5; ++k) send[s * kRecW + k] = dep[i * 9 + k];
          for (int k = 0; k < 3; ++k)
            send[s * kRecW + 5 + k] = dep[i * 9 + 6 + k];
        }
      }
      dep.clear();

      std::vector<double> recv;
      if (world_ == 1) {
        if (m == 0) break;
        recv = std::move(send);
      } else {
        std::vector<int64_t> flat((int64_t)world_ * world_, 0);
        for (int r = 0; r < world_; ++r)
          flat[(int64_t)rank_ * world_ + r] = scounts[r];
        comm_->allreduce_sum(flat.data(), (int64_t)world_ * world_);
        int64_t total = 0;
        for (int64_t c : flat) total += c;
        if (total == 0) break;
        std::vector<int64_t> sc(world_);
        for (int r = 0; r < world_; ++r) sc[r] = scounts[r] * kRecW;
        recv = comm_->alltoallv(send.data(), sc);
      }
      const int64_t nr = (int64_t)recv.size() / kRecW;
      if (nr && ovr.empty()) ovr.assign(n_ * 3, 0.0);
      for (int64_t i = 0; i < nr; ++i) {
        const int64_t g = (int64_t)recv[i * kRecW];
        res_[g] = 1;
        esc_[g] = 0;
        pos_[g * 3] = recv[i * kRecW + 1];
        pos_[g * 3 + 1] = recv[i * kRecW + 2];
        pos_[g * 3 + 2] = recv[i * kRecW + 3];
        elem_[g] = dec_.g2l[(int64_t)recv[i * kRecW + 4]];
        ovr[g * 3] = recv[i * kRecW + 5];
        ovr[g * 3 + 1] = recv[i * kRecW + 6];
        ovr[g * 3 + 2] = recv[i * kRecW + 7];
        list.push_back((int32_t)g);
      }
      if (round == max_rounds_ - 1)
        throw std::runtime_error("partitioned step did not converge in " +
                                 std::to_string(max_rounds_) +
                                 " handoff rounds");
    }
    stats_.moves++;
  }

  std::vector<double> flux_global() override {
    const std::vector<double> local = eng_->flux();
    const int64_t nl = dec_.sub.local.nelems;
    const int64_t slabs = (int64_t)nscores_ * ngroups_;
    std::vector<double> out(slabs * nelems_global_, 0.0);
    for (int64_t s = 0; s < slabs; ++s)
      for (int64_t t = 0; t < nl; ++t)
        out[s * nelems_global_ + dec_.sub.elem_l2g[t]] +=
            local[s * nl + t];
    if (world_ > 1)
      comm_->allreduce_sum(out.data(), (int64_t)out.size());
    return out;
  }

  int64_t resident() const override {
    int64_t c = 0;
    for (uint8_t v : res_) c += v;
    return c;
  }

  const EngineStats &stats() const override {
    const EngineStats &inner = eng_->stats();
    stats_.lost_particles = inner.lost_particles;
    return stats_;
  }

  void synchronize() override {}

  std::vector<uint8_t> resident_mask() const override { return res_; }
  std::vector<double> positions() const override { return pos_; }
  std::vector<int32_t> elem_ids() const override { return elem_; }

public:
  std::function<int32_t(Vec3, double, bool *)> full_locate_;
  int max_rounds_ = 64;

private:
  void check_n(int64_t n) const {
    if (n != n_) throw std::runtime_error("global particle count mismatch");
  }

  int64_t n_;
  Comm *comm_;
  int rank_, world_;
  int ngroups_, nscores_ = 1;
  int64_t nelems_global_;
  Decomp dec_;
  std::unique_ptr<Engine> eng_;
  double loc_tol_ = 0.0;
  mutable EngineStats stats_;
  std::vector<double> pos_;
  std::vector<int32_t> elem_;
  std::vector<uint8_t> res_, esc_;
};

} // namespace

std::unique_ptr<PartitionedEngine> make_partitioned_engine(
    const Mesh &full, int64_t n_global, Comm *comm, int rank, int world,
    const std::string &device, int ngroups, int nscores,
    const int32_t *owners, int ghost_rings) {
  // the rare global-resolve path keeps a host copy of the full mesh
  auto full_copy = std::make_shared<Mesh>(full);
  auto locate = [full_copy](Vec3 q, double tol, bool *lo) {
    return full_copy->locate(q, tol, lo);
  };
  if (device != "cpu") {
    int ordinal = 0;
    if (device.rfind("cuda:", 0) == 0) ordinal = atoi(device.c_str() + 5);
    else if (!device.empty() && device != "auto") ordinal = atoi(device.c_str());
    int count = 0;
    if (hipGetDeviceCount(&count) == hipSuccess && count > ordinal) {
      auto e = std::make_unique<GpuPartitionedEngine>(
          full, n_global, comm, rank, world, ordinal, ngroups, nscores,
          owners, ghost_rings);
      e->full_locate_ = locate;
      return e;
    }
    (void)hipGetLastError();
    if (device != "auto")
      throw std::runtime_error("PartitionedEngine: HIP device unavailable");
  }
  auto e = std::make_unique<CpuPartitionedEngine>(full, n_global, comm, rank,
                                                  world, ngroups, nscores,
                                                  owners, ghost_rings);
  e->full_locate_ = locate;
  return e;
}

} // namespace pumitally
