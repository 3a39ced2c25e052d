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

// Record layout (runtime width rec_w = 9 + carry_grp + nscores):
//   [0]=gid [1..3]=pos [4]=target_gid [5..7]=dest [8]=weight
//   [9]=group (iff ngroups>1) [9+cg ..]=nscores response multipliers
// The destination travels because reflective/periodic restarts inside
// the walk MUTATE it (walk.h out_dest); weight/group/responses travel so
// the coupled-host path (step_local) works -- the receiver never saw
// the sender's input arrays.  In global-array mode the receiver scatters
// the carried values into its own global-indexed buffers (identical
// values, so both modes share one code path).
// dep entries append the destination OWNER rank at column [rec_w].
constexpr int kPBlock = 256;
constexpr int kMaxChunks = 8; // step() copy/walk pipeline depth

inline int pgrid(int64_t n) {
  int64_t b = (n + kPBlock - 1) / kPBlock;
  if (b > 2048) b = 2048;
  if (b < 1) b = 1;
  return (int)b;
}

// Streaming one-pass kernels (prepare/gather/collect/unpack) want one
// item per thread: the grid-stride cap that suits the walk launches
// leaves them latency-bound (~0.24 ms per 1.25M-item prepare measured
// at 2048 blocks).
inline int pgrid_flat(int64_t n) {
  int64_t b = (n + kPBlock - 1) / kPBlock;
  if (b > 65535) b = 65535;
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
    // claim STRICT hits only: a loose (tol*1e4) hit on the SUBMESH can
    // steal a point whose true element is absent from this rank's ghost
    // ring (found at ~7e-7 depth by tools/part_world2_soak); such points
    // are resolved on the host against the FULL mesh -- the same
    // decision the replicated oracle makes -- so the starting element
    // (and hence the first tally sliver) matches bitwise.
    if (le >= 0 && !lo && lowner[le] == myrank) {
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

// dep entry (rec_w doubles + owner): [gid, ox, oy, oz, target_gid,
// dx, dy, dz, weight, t, prev_gid (, grp)(, resp x nscores)] [owner].
// (ox,oy,oz) is the walk's CURRENT wrap-segment origin and t its
// progress at the cut crossing; prev_gid is the global id of the
// element exited from.  Carrying the t-parametrization (instead of the
// crossing point) lets the receiving rank resume with the sender's
// exact fp state, so partitioned flux attribution is elementwise
// identical to the replicated engine (walk.h walk_segment doc).
// Fresh entries (phase-A reroutes, host ejects) use t=0, prev=-1.
// Operates on the gid range [g_lo, g_hi): step() pipelines chunks of the
// batch so chunk c's prepare+walk overlap chunk c+1's H2D copies.  The
// chunk's walk list lives in list[g_lo ...] with its own counter
// (nwalk_ctr), so segments never collide.
__device__ __forceinline__ void dep_fill_tail(
    double *e, int rec_w, double weight, bool carry_grp, uint16_t grp,
    const double *resp, int nscores) {
  e[8] = weight;
  int at = 11; // [9]=t and [10]=prev_gid are written by the caller
  if (carry_grp) e[at++] = (double)grp;
  for (int k = 0; k < nscores; ++k) e[at + k] = resp ? resp[k] : 1.0;
  (void)rec_w;
}

__global__ void k_part_prepare(
    int64_t g_lo, int64_t g_hi, uint8_t *res /* read+clear, no restrict */,
    const uint8_t *__restrict__ esc, const int8_t *__restrict__ fly,
    const double *__restrict__ orig, double *__restrict__ pos,
    int32_t *__restrict__ elem, const Plane *__restrict__ planes,
    GridView grid, const int32_t *__restrict__ lowner,
    const int32_t *__restrict__ l2g, int myrank, double tol,
    const double *__restrict__ dest, const double *__restrict__ w,
    const uint16_t *__restrict__ grp, const double *__restrict__ resp,
    int nscores, bool carry_grp, int rec_w, int32_t *__restrict__ list,
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
        const int32_t le0 = grid_locate(grid, planes, q, tol, &lo);
        // loose submesh hits are NOT trusted (see k_part_localize): the
        // host resolves them on the full mesh, bitwise-matching the
        // replicated oracle's relocation
        const int32_t le = lo ? -1 : le0;
        if (le >= 0) {
          if (lowner[le] == myrank) {
            elem[g] = le;
            pos[g * 3] = q.x;
            pos[g * 3 + 1] = q.y;
            pos[g * 3 + 2] = q.z;
          } else {
            // resampled into a ghost element: reroute to its owner
            const unsigned long long k = atomicAdd(&ctr[1], 1ull);
            double *e = dep + k * (rec_w + 1);
            e[0] = (double)g;
            e[1] = q.x;
            e[2] = q.y;
            e[3] = q.z;
            e[4] = (double)l2g[le];
            e[5] = dest[g * 3];
            e[6] = dest[g * 3 + 1];
            e[7] = dest[g * 3 + 2];
            e[9] = 0.0;   // fresh walk: no resume state
            e[10] = -1.0;
            dep_fill_tail(e, rec_w, w[g], carry_grp,
                          grp ? grp[g] : (uint16_t)0,
                          resp ? resp + (int64_t)g * nscores : nullptr,
                          nscores);
            e[rec_w] = (double)lowner[le];
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
                              double *__restrict__ wresp,
                              const double *__restrict__ t0 = nullptr,
                              const int32_t *__restrict__ prev = nullptr,
                              double *__restrict__ wt0 = nullptr,
                              int32_t *__restrict__ wprev = nullptr) {
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
    if (wt0) wt0[j] = t0 ? t0[g] : 0.0;
    if (wprev) wprev[j] = prev ? prev[g] : -1;
  }
}

__global__ void k_part_collect(const int32_t *__restrict__ list, int64_t m,
                               const double *__restrict__ wout_pos,
                               const int32_t *__restrict__ wout_elem,
                               const int8_t *__restrict__ wstatus,
                               const double *__restrict__ wout_dest,
                               const double *__restrict__ wout_o,
                               const double *__restrict__ wout_t,
                               const int32_t *__restrict__ wout_prev,
                               const int32_t *__restrict__ l2g,
                               const double *__restrict__ ww,
                               const uint16_t *__restrict__ wgrp,
                               const double *__restrict__ wresp,
                               int nscores, bool carry_grp, int rec_w,
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
      double *e = dep + d * (rec_w + 1);
      e[0] = (double)g;
      // resume state: wrap-segment origin, progress t, exited-from elem
      e[1] = wout_o[j * 3];
      e[2] = wout_o[j * 3 + 1];
      e[3] = wout_o[j * 3 + 2];
      e[4] = (double)fgid[k];
      e[5] = wout_dest[j * 3];
      e[6] = wout_dest[j * 3 + 1];
      e[7] = wout_dest[j * 3 + 2];
      e[9] = wout_t[j];
      e[10] = wout_prev[j] >= 0 ? (double)l2g[wout_prev[j]] : -1.0;
      dep_fill_tail(e, rec_w, ww[j], carry_grp,
                    wgrp ? wgrp[j] : (uint16_t)0,
                    wresp ? wresp + (int64_t)j * nscores : nullptr, nscores);
      e[rec_w] = (double)fowner[k];
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
                             int rec_w,
                             unsigned long long *__restrict__ dcnt) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride)
    atomicAdd(&dcnt[(int)dep[i * (rec_w + 1) + rec_w]], 1ull);
}

__global__ void k_part_pack(const double *__restrict__ dep, int64_t m,
                            int rec_w, const int64_t *__restrict__ offs,
                            unsigned long long *__restrict__ cur,
                            double *__restrict__ send) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride) {
    const double *e = dep + i * (rec_w + 1);
    const int o = (int)e[rec_w];
    const int64_t s = offs[o] + (int64_t)atomicAdd(&cur[o], 1ull);
    for (int k = 0; k < rec_w; ++k) send[s * rec_w + k] = e[k];
  }
}

__global__ void k_part_unpack(const double *__restrict__ recv, int64_t m,
                              int rec_w, bool carry_grp, int nscores,
                              const int32_t *__restrict__ g2l,
                              double *__restrict__ pos,
                              int32_t *__restrict__ elem,
                              uint8_t *__restrict__ res,
                              uint8_t *__restrict__ esc,
                              double *__restrict__ dest_ovr,
                              double *__restrict__ w_out,
                              uint16_t *__restrict__ grp_out,
                              double *__restrict__ resp_out,
                              double *__restrict__ t0_out,
                              int32_t *__restrict__ prev_out,
                              int32_t *__restrict__ list,
                              unsigned long long *__restrict__ ctr) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride) {
    const double *e = recv + i * rec_w;
    const int64_t g = (int64_t)e[0];
    res[g] = 1;
    esc[g] = 0;
    pos[g * 3] = e[1];
    pos[g * 3 + 1] = e[2];
    pos[g * 3 + 2] = e[3];
    elem[g] = g2l[(int64_t)e[4]];
    dest_ovr[g * 3] = e[5];
    dest_ovr[g * 3 + 1] = e[6];
    dest_ovr[g * 3 + 2] = e[7];
    // scatter the carried step inputs so later rounds (and the shared
    // gather-by-gid path) see them regardless of which mode sent them
    w_out[g] = e[8];
    t0_out[g] = e[9];
    const int64_t pg = (int64_t)e[10];
    prev_out[g] = pg >= 0 ? g2l[pg] : -1; // -1: outside submesh (no rings)
    int at = 11;
    if (carry_grp && grp_out) grp_out[g] = (uint16_t)e[at];
    at += carry_grp ? 1 : 0;
    if (resp_out)
      for (int k = 0; k < nscores; ++k)
        resp_out[(int64_t)g * nscores + k] = e[at + k];
    list[wave_append(&ctr[0])] = (int32_t)g;
  }
}

// apply host-resolved out-of-mesh relocations: particle stays resident
// here with elem=-1 at its requested origin
// apply host-resolved full-mesh claims from localize(): particle g
// becomes resident here in local element lel[i] at its origin (already
// on device in d_dest_)
__global__ void k_part_apply_claims(const int32_t *__restrict__ gids,
                                    const int32_t *__restrict__ lels,
                                    int64_t m,
                                    const double *__restrict__ orig,
                                    double *__restrict__ pos,
                                    int32_t *__restrict__ elem,
                                    uint8_t *__restrict__ res,
                                    uint8_t *__restrict__ esc) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < m; i += stride) {
    const int64_t g = gids[i];
    res[g] = 1;
    esc[g] = 0;
    elem[g] = lels[i];
    pos[g * 3] = orig[g * 3];
    pos[g * 3 + 1] = orig[g * 3 + 1];
    pos[g * 3 + 2] = orig[g * 3 + 2];
  }
}

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

// compact the resident mask into an ordered gid list (coupled-host
// frame); wave_append keeps rough gid order
__global__ void k_part_compact(const uint8_t *__restrict__ res, int64_t n,
                               int32_t *__restrict__ frame,
                               unsigned long long *__restrict__ ctr) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t g = blockIdx.x * blockDim.x + threadIdx.x; g < n; g += stride)
    if (res[g]) frame[wave_append(ctr)] = (int32_t)g;
}

// coupled-host round 0: inputs indexed by frame position j, state by the
// frame's gid; the gather into the walk scratch is fused (there is no
// gid-indexed input array to gather from)
__global__ void k_part_prepare_local(
    int64_t nloc, const int32_t *__restrict__ frame, uint8_t *res,
    const uint8_t *__restrict__ esc, const int8_t *__restrict__ fly,
    const double *__restrict__ orig, double *__restrict__ pos,
    int32_t *__restrict__ elem, const Plane *__restrict__ planes,
    GridView grid, const int32_t *__restrict__ lowner,
    const int32_t *__restrict__ l2g, int myrank, double tol,
    const double *__restrict__ dest, const double *__restrict__ w,
    const uint16_t *__restrict__ grp, const double *__restrict__ resp,
    int nscores, bool carry_grp, int rec_w, int32_t *__restrict__ list,
    double *__restrict__ wpos, double *__restrict__ wdest,
    int32_t *__restrict__ welem, double *__restrict__ ww,
    uint16_t *__restrict__ wgrp, double *__restrict__ wresp,
    double *__restrict__ dep, int32_t *__restrict__ eject,
    unsigned long long *__restrict__ ctr,
    unsigned long long *__restrict__ nwalk_ctr) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = blockIdx.x * blockDim.x + threadIdx.x; j < nloc;
       j += stride) {
    const int64_t g = frame[j];
    if (!res[g] || !fly[j]) continue;
    Vec3 p{pos[g * 3], pos[g * 3 + 1], pos[g * 3 + 2]};
    if (orig != nullptr && !esc[g]) {
      const Vec3 q{orig[j * 3], orig[j * 3 + 1], orig[j * 3 + 2]};
      if (q.x != p.x || q.y != p.y || q.z != p.z) {
        atomicAdd(&ctr[3], 1ull);
        bool lo = false;
        const int32_t le0 = grid_locate(grid, planes, q, tol, &lo);
        const int32_t le = lo ? -1 : le0; // see k_part_prepare
        if (le >= 0) {
          if (lowner[le] == myrank) {
            elem[g] = le;
            pos[g * 3] = q.x;
            pos[g * 3 + 1] = q.y;
            pos[g * 3 + 2] = q.z;
            p = q;
          } else {
            const unsigned long long k = atomicAdd(&ctr[1], 1ull);
            double *e = dep + k * (rec_w + 1);
            e[0] = (double)g;
            e[1] = q.x;
            e[2] = q.y;
            e[3] = q.z;
            e[4] = (double)l2g[le];
            e[5] = dest[j * 3];
            e[6] = dest[j * 3 + 1];
            e[7] = dest[j * 3 + 2];
            e[9] = 0.0;   // fresh walk: no resume state
            e[10] = -1.0;
            dep_fill_tail(e, rec_w, w[j], carry_grp,
                          grp ? grp[j] : (uint16_t)0,
                          resp ? resp + j * nscores : nullptr, nscores);
            e[rec_w] = (double)lowner[le];
            res[g] = 0;
            continue;
          }
        } else {
          const unsigned long long k = atomicAdd(&ctr[2], 1ull);
          eject[k] = (int32_t)j; // frame index: host inputs are by j
          continue;
        }
      }
    }
    if (elem[g] < 0) continue;
    const int64_t slot = wave_append(nwalk_ctr);
    list[slot] = (int32_t)g;
    wpos[slot * 3] = p.x;
    wpos[slot * 3 + 1] = p.y;
    wpos[slot * 3 + 2] = p.z;
    wdest[slot * 3] = dest[j * 3];
    wdest[slot * 3 + 1] = dest[j * 3 + 1];
    wdest[slot * 3 + 2] = dest[j * 3 + 2];
    welem[slot] = elem[g];
    ww[slot] = w[j];
    if (wgrp) wgrp[slot] = grp[j];
    if (wresp)
      for (int k = 0; k < nscores; ++k)
        wresp[slot * nscores + k] = resp[j * nscores + k];
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

    // record widths (see the layout comment at the top)
    carry_grp_ = ngroups_ > 1;
    rec_w_ = 11 + (carry_grp_ ? 1 : 0) + nscores_;

    // work buffers
    d_list_ = pdmalloc<int32_t>(n_);
    d_dep_ = pdmalloc<double>(n_ * (rec_w_ + 1));
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
    // bitwise handoff-resume state (record layout comment above)
    d_t0_ = pdmalloc<double>(n_);
    d_prev_ = pdmalloc<int32_t>(n_);
    d_wt0_ = pdmalloc<double>(n_);
    d_wprev_ = pdmalloc<int32_t>(n_);
    d_wout_o_ = pdmalloc<double>(n_ * 3);
    d_wout_t_ = pdmalloc<double>(n_);
    d_wout_prev_ = pdmalloc<int32_t>(n_);
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
          (void *)d_resp_, (void *)d_wresp_, (void *)d_frame_,
          (void *)d_ldest_, (void *)d_lw_, (void *)d_lorig_,
          (void *)d_lresp_, (void *)d_lfly_, (void *)d_lgrp_,
          (void *)d_t0_, (void *)d_prev_, (void *)d_wt0_,
          (void *)d_wprev_, (void *)d_wout_o_, (void *)d_wout_t_,
          (void *)d_wout_prev_})
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
    k_part_localize<<<pgrid_flat(n_), kPBlock>>>(
        dmesh_.planes, dmesh_.grid, d_lowner_, rank_, d_dest_, n_, loc_tol_,
        d_pos_, d_elem_, d_res_, d_esc_, d_claim, &d_ctr_[4]);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipDeviceSynchronize());
    // claims are disjoint (only the owner claims), so bitwise OR of the
    // claim masks == integer sum of the words
    std::vector<int64_t> words(nwords);
    PT_HIP_CHECK(hipMemcpy(words.data(), d_claim, nwords * 8,
                           hipMemcpyDeviceToHost));
    if (world_ > 1) comm_->allreduce_sum(words.data(), nwords);
    // Unclaimed = outside the mesh OR only loosely localizable on a
    // submesh (true element absent from the ghost ring, or the point is
    // within tolerance of a cut face).  Resolve against the FULL mesh on
    // the host -- every rank computes the same answer deterministically,
    // so the owner claims without extra communication, and the chosen
    // element matches the replicated oracle's bitwise.
    if (full_locate_) {
      std::vector<int32_t> cg, cl;
      for (int64_t g = 0; g < n_; ++g) {
        if ((words[g >> 6] >> (g & 63)) & 1) continue;
        const Vec3 q{origins[g * 3], origins[g * 3 + 1], origins[g * 3 + 2]};
        bool lo = false;
        const int32_t ge = full_locate_(q, loc_tol_, &lo);
        if (ge < 0) continue; // truly outside: rank 0 parks it below
        if (lo) stats_.loose_localizations++;
        words[g >> 6] |= (int64_t)(1ull << (g & 63));
        if (dec_.owners[ge] == rank_) {
          cg.push_back((int32_t)g);
          cl.push_back(dec_.g2l[ge]);
        }
      }
      if (!cg.empty()) {
        int32_t *d_cg = pdmalloc<int32_t>((int64_t)cg.size());
        int32_t *d_cl = pdmalloc<int32_t>((int64_t)cl.size());
        PT_HIP_CHECK(hipMemcpy(d_cg, cg.data(), cg.size() * 4,
                               hipMemcpyHostToDevice));
        PT_HIP_CHECK(hipMemcpy(d_cl, cl.data(), cl.size() * 4,
                               hipMemcpyHostToDevice));
        k_part_apply_claims<<<pgrid((int64_t)cg.size()), kPBlock>>>(
            d_cg, d_cl, (int64_t)cg.size(), d_dest_, d_pos_, d_elem_,
            d_res_, d_esc_);
        PT_HIP_CHECK(hipGetLastError());
        PT_HIP_CHECK(hipDeviceSynchronize());
        PT_HIP_CHECK(hipFree(d_cg));
        PT_HIP_CHECK(hipFree(d_cl));
      }
    }
    PT_HIP_CHECK(hipMemcpy(d_claim, words.data(), nwords * 8,
                           hipMemcpyHostToDevice));
    if (rank_ == 0) {
      k_part_claim_rest<<<pgrid_flat(n_), kPBlock>>>(d_claim, d_dest_, n_, d_pos_,
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
      k_part_prepare<<<pgrid_flat(hi - lo), kPBlock, 0, cs_>>>(
          lo, hi, d_res_, d_esc_, d_fly_, origin ? d_orig_ : nullptr,
          d_pos_, d_elem_, dmesh_.planes, dmesh_.grid, d_lowner_, d_l2g_,
          rank_, loc_tol_, d_dest_, d_w_, groups ? d_grp_ : nullptr,
          responses ? d_resp_ : nullptr, nscores_, carry_grp_, rec_w_,
          d_list_, d_dep_, d_eject_, d_ctr_, &d_ctr_[ctr_chunk0_ + c]);
      PT_HIP_CHECK(hipGetLastError());
      PT_HIP_CHECK(hipStreamSynchronize(cs_));
      unsigned long long cw = 0;
      PT_HIP_CHECK(hipMemcpy(&cw, &d_ctr_[ctr_chunk0_ + c], 8,
                             hipMemcpyDeviceToHost));
      if (cw == 0) continue;
      nwalk += (int64_t)cw;
      k_part_gather<<<pgrid_flat((int64_t)cw), kPBlock, 0, cs_>>>(
          d_list_ + lo, (int64_t)cw, d_pos_, d_elem_, d_dest_, d_dest_ovr_,
          /*use_ovr=*/false, d_w_, d_grp_, d_resp_, nscores_, d_wpos_,
          d_wdest_, d_welem_, d_ww_, groups ? d_wgrp_ : nullptr,
          responses ? d_wresp_ : nullptr);
      PT_HIP_CHECK(hipGetLastError());
      eng_->walk_raw_device((int64_t)cw, d_wpos_, d_wdest_, d_welem_,
                            d_ww_, d_wout_pos_, d_wout_elem_, d_wstatus_,
                            groups ? d_wgrp_ : nullptr,
                            responses ? d_wresp_ : nullptr, d_wout_dest_,
                            nullptr, nullptr, d_wout_o_, d_wout_t_,
                            d_wout_prev_);
      k_part_collect<<<pgrid_flat((int64_t)cw), kPBlock, 0, cs_>>>(
          d_list_ + lo, (int64_t)cw, d_wout_pos_, d_wout_elem_, d_wstatus_,
          d_wout_dest_, d_wout_o_, d_wout_t_, d_wout_prev_, d_l2g_, d_ww_,
          groups ? d_wgrp_ : nullptr,
          responses ? d_wresp_ : nullptr, nscores_, carry_grp_, rec_w_,
          d_pos_, d_elem_, d_esc_, d_res_, d_fgid_,
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
    if (hctr[2] > 0)
      host_resolve_ejects((int64_t)hctr[2], origin, dest, weights, groups,
                          responses);
    stats_.relocated += (int64_t)hctr[3];
    stats_.loose_localizations += (int64_t)hctr[4];

    run_exchange_rounds(groups != nullptr, responses != nullptr);
    stats_.moves++;
  }

  // rounds 1+ of a step: walk lists come from exchanged records (round
  // 0's walks already ran); shared by step() and step_local().
  void run_exchange_rounds(bool groups_used, bool resp_used) {
    int64_t nwalk_r = 0;
    for (int round = 1; round <= max_rounds_; ++round) {
      if (nwalk_r > 0) {
        k_part_gather<<<pgrid_flat(nwalk_r), kPBlock, 0, cs_>>>(
            d_list_, nwalk_r, d_pos_, d_elem_, d_dest_, d_dest_ovr_,
            /*use_ovr=*/true, d_w_, d_grp_, d_resp_,
            nscores_, d_wpos_, d_wdest_, d_welem_, d_ww_,
            groups_used ? d_wgrp_ : nullptr, resp_used ? d_wresp_ : nullptr,
            d_t0_, d_prev_, d_wt0_, d_wprev_);
        PT_HIP_CHECK(hipGetLastError());
        eng_->walk_raw_device(nwalk_r, d_wpos_, d_wdest_, d_welem_, d_ww_,
                              d_wout_pos_, d_wout_elem_, d_wstatus_,
                              groups_used ? d_wgrp_ : nullptr,
                              resp_used ? d_wresp_ : nullptr, d_wout_dest_,
                              d_wt0_, d_wprev_, d_wout_o_, d_wout_t_,
                              d_wout_prev_);
        k_part_collect<<<pgrid_flat(nwalk_r), kPBlock, 0, cs_>>>(
            d_list_, nwalk_r, d_wout_pos_, d_wout_elem_, d_wstatus_,
            d_wout_dest_, d_wout_o_, d_wout_t_, d_wout_prev_, d_l2g_, d_ww_,
            groups_used ? d_wgrp_ : nullptr,
            resp_used ? d_wresp_ : nullptr, nscores_, carry_grp_, rec_w_,
            d_pos_, d_elem_, d_esc_, d_res_, d_fgid_,
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
        k_part_count<<<pgrid_flat(m), kPBlock, 0, cs_>>>(d_dep_, m, rec_w_,
                                                       &d_ctr_[8]);
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
        ensure_cap(&d_send_, &cap_send_, m * rec_w_);
        PT_HIP_CHECK(hipMemcpy(d_offs_, offs.data(), world_ * 8,
                               hipMemcpyHostToDevice));
        PT_HIP_CHECK(hipMemsetAsync(&d_ctr_[8 + world_], 0, world_ * 8,
                                    cs_));
        k_part_pack<<<pgrid_flat(m), kPBlock, 0, cs_>>>(
            d_dep_, m, rec_w_, d_offs_, &d_ctr_[8 + world_], d_send_);
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
          sc[r] = scounts[r] * rec_w_;
          rc[r] = rcounts[r] * rec_w_;
        }
        for (int64_t c : rcounts) nrecv += c;
        if (comm_->has_device_collectives()) {
          double *d_recv = nullptr;
          comm_->alltoallv_device(d_send_, sc, rc, &d_recv);
          recv_ptr = d_recv;
        } else {
          // host-staged exchange (TCP fallback): D2H -> alltoallv -> H2D
          std::vector<double> hsend(m * rec_w_);
          if (m)
            PT_HIP_CHECK(hipMemcpy(hsend.data(), d_send_, m * rec_w_ * 8,
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
        k_part_unpack<<<pgrid_flat(nrecv), kPBlock, 0, cs_>>>(
            recv_ptr, nrecv, rec_w_, carry_grp_, nscores_, d_g2l_, d_pos_,
            d_elem_, d_res_, d_esc_, d_dest_ovr_, d_w_, d_grp_, d_resp_,
            d_t0_, d_prev_, d_list_, d_ctr_);
        PT_HIP_CHECK(hipGetLastError());
      }
      PT_HIP_CHECK(hipStreamSynchronize(cs_));
      nwalk_r = nrecv;
      if (round == max_rounds_)
        throw std::runtime_error("partitioned step did not converge in " +
                                 std::to_string(max_rounds_) +
                                 " handoff rounds");
    }
  }

  std::vector<int64_t> resident_list() override {
    PT_HIP_CHECK(hipSetDevice(device_));
    eng_->synchronize();
    if (!d_frame_) d_frame_ = pdmalloc<int32_t>(n_);
    PT_HIP_CHECK(hipMemsetAsync(&d_ctr_[5], 0, 8, cs_));
    k_part_compact<<<pgrid_flat(n_), kPBlock, 0, cs_>>>(d_res_, n_, d_frame_,
                                                       &d_ctr_[5]);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipStreamSynchronize(cs_));
    unsigned long long nf = 0;
    PT_HIP_CHECK(hipMemcpy(&nf, &d_ctr_[5], 8, hipMemcpyDeviceToHost));
    n_frame_ = (int64_t)nf;
    std::vector<int32_t> f32(n_frame_);
    if (n_frame_)
      PT_HIP_CHECK(hipMemcpy(f32.data(), d_frame_, n_frame_ * 4,
                             hipMemcpyDeviceToHost));
    h_frame_.assign(f32.begin(), f32.end());
    return h_frame_;
  }

  void step_local(const double *dest, const int8_t *flying,
                  const double *weights, int64_t n_local,
                  const double *origin, const uint16_t *groups,
                  const double *responses) override {
    PT_HIP_CHECK(hipSetDevice(device_));
    if (n_frame_ < 0)
      throw std::runtime_error(
          "step_local: call resident_list() first (it defines the input "
          "order this call consumes)");
    if (n_local != n_frame_)
      throw std::runtime_error("step_local: n_local " +
                               std::to_string(n_local) +
                               " != resident_list size " +
                               std::to_string(n_frame_));
    if (groups && ngroups_ <= 1)
      throw std::runtime_error("groups passed but ngroups == 1");
    eng_->synchronize();
    if (!d_ldest_) {
      d_ldest_ = pdmalloc<double>(n_ * 3);
      d_lw_ = pdmalloc<double>(n_);
      d_lfly_ = pdmalloc<int8_t>(n_);
    }
    if (origin && !d_lorig_) d_lorig_ = pdmalloc<double>(n_ * 3);
    if (groups && !d_lgrp_) d_lgrp_ = pdmalloc<uint16_t>(n_);
    if (responses && !d_lresp_) d_lresp_ = pdmalloc<double>(n_ * nscores_);
    if (responses && !d_resp_) {
      d_resp_ = pdmalloc<double>(n_ * nscores_);
      d_wresp_ = pdmalloc<double>(n_ * nscores_);
    }
    if (groups && !d_grp_) {
      d_grp_ = pdmalloc<uint16_t>(n_);
      d_wgrp_ = pdmalloc<uint16_t>(n_);
    }
    if (n_local) {
      PT_HIP_CHECK(hipMemcpy(d_ldest_, dest, n_local * 24,
                             hipMemcpyHostToDevice));
      PT_HIP_CHECK(hipMemcpy(d_lfly_, flying, n_local,
                             hipMemcpyHostToDevice));
      PT_HIP_CHECK(hipMemcpy(d_lw_, weights, n_local * 8,
                             hipMemcpyHostToDevice));
      if (origin)
        PT_HIP_CHECK(hipMemcpy(d_lorig_, origin, n_local * 24,
                               hipMemcpyHostToDevice));
      if (groups)
        PT_HIP_CHECK(hipMemcpy(d_lgrp_, groups, n_local * 2,
                               hipMemcpyHostToDevice));
      if (responses)
        PT_HIP_CHECK(hipMemcpy(d_lresp_, responses,
                               n_local * nscores_ * 8,
                               hipMemcpyHostToDevice));
    }
    PT_HIP_CHECK(hipMemsetAsync(d_ctr_, 0,
                                (ctr_chunk0_ + kMaxChunks) * 8, cs_));
    if (n_local) {
      k_part_prepare_local<<<pgrid_flat(n_local), kPBlock, 0, cs_>>>(
          n_local, d_frame_, d_res_, d_esc_, d_lfly_,
          origin ? d_lorig_ : nullptr, d_pos_, d_elem_, dmesh_.planes,
          dmesh_.grid, d_lowner_, d_l2g_, rank_, loc_tol_, d_ldest_, d_lw_,
          groups ? d_lgrp_ : nullptr, responses ? d_lresp_ : nullptr,
          nscores_, carry_grp_, rec_w_, d_list_, d_wpos_, d_wdest_,
          d_welem_, d_ww_, groups ? d_wgrp_ : nullptr,
          responses ? d_wresp_ : nullptr, d_dep_, d_eject_, d_ctr_,
          &d_ctr_[ctr_chunk0_]);
      PT_HIP_CHECK(hipGetLastError());
    }
    PT_HIP_CHECK(hipStreamSynchronize(cs_));
    unsigned long long hctr[5] = {0, 0, 0, 0, 0}, cw = 0;
    PT_HIP_CHECK(hipMemcpy(hctr, d_ctr_, 5 * 8, hipMemcpyDeviceToHost));
    PT_HIP_CHECK(hipMemcpy(&cw, &d_ctr_[ctr_chunk0_], 8,
                           hipMemcpyDeviceToHost));
    if (hctr[2] > 0)
      host_resolve_ejects_local((int64_t)hctr[2], origin, dest, weights,
                                groups, responses);
    stats_.relocated += (int64_t)hctr[3];
    stats_.loose_localizations += (int64_t)hctr[4];
    if (cw > 0) {
      eng_->walk_raw_device((int64_t)cw, d_wpos_, d_wdest_, d_welem_, d_ww_,
                            d_wout_pos_, d_wout_elem_, d_wstatus_,
                            groups ? d_wgrp_ : nullptr,
                            responses ? d_wresp_ : nullptr, d_wout_dest_,
                            nullptr, nullptr, d_wout_o_, d_wout_t_,
                            d_wout_prev_);
      k_part_collect<<<pgrid_flat((int64_t)cw), kPBlock, 0, cs_>>>(
          d_list_, (int64_t)cw, d_wout_pos_, d_wout_elem_, d_wstatus_,
          d_wout_dest_, d_wout_o_, d_wout_t_, d_wout_prev_, d_l2g_, d_ww_,
          groups ? d_wgrp_ : nullptr,
          responses ? d_wresp_ : nullptr, nscores_, carry_grp_, rec_w_,
          d_pos_, d_elem_, d_esc_, d_res_, d_fgid_, d_fowner_, d_dep_,
          d_ctr_);
      PT_HIP_CHECK(hipGetLastError());
      PT_HIP_CHECK(hipStreamSynchronize(cs_));
    }
    run_exchange_rounds(groups != nullptr, responses != nullptr);
    n_frame_ = -1; // residency changed; a new snapshot is required
    stats_.moves++;
  }

  // step_local's eject resolution: inputs are frame-indexed
  void host_resolve_ejects_local(int64_t ne, const double *origin,
                                 const double *dest, const double *weights,
                                 const uint16_t *groups,
                                 const double *responses) {
    std::vector<int32_t> js(ne);
    PT_HIP_CHECK(
        hipMemcpy(js.data(), d_eject_, ne * 4, hipMemcpyDeviceToHost));
    const int dw = rec_w_ + 1;
    std::vector<double> depx;
    std::vector<int32_t> out_j;
    for (int64_t i = 0; i < ne; ++i) {
      const int64_t j = js[i];
      const int64_t g = h_frame_[j];
      const Vec3 q{origin[j * 3], origin[j * 3 + 1], origin[j * 3 + 2]};
      bool lo = false;
      const int32_t ge = full_locate_ ? full_locate_(q, loc_tol_, &lo) : -1;
      if (lo) stats_.loose_localizations++;
      if (ge >= 0) {
        std::vector<double> e(dw, 0.0);
        e[0] = (double)g;
        e[1] = q.x;
        e[2] = q.y;
        e[3] = q.z;
        e[4] = (double)ge;
        e[5] = dest[j * 3];
        e[6] = dest[j * 3 + 1];
        e[7] = dest[j * 3 + 2];
        e[8] = weights[j];
        e[9] = 0.0; // fresh walk: no resume state
        e[10] = -1.0;
        int at = 11;
        if (carry_grp_) e[at++] = groups ? (double)groups[j] : 0.0;
        for (int k = 0; k < nscores_; ++k)
          e[at + k] = responses ? responses[j * nscores_ + k] : 1.0;
        e[rec_w_] = (double)dec_.owners[ge];
        depx.insert(depx.end(), e.begin(), e.end());
        // the particle left this rank
        const uint8_t zero = 0;
        PT_HIP_CHECK(
            hipMemcpy(d_res_ + g, &zero, 1, hipMemcpyHostToDevice));
      } else {
        out_j.push_back((int32_t)j);
      }
    }
    if (!depx.empty()) {
      unsigned long long ndep = 0;
      PT_HIP_CHECK(hipMemcpy(&ndep, &d_ctr_[1], 8, hipMemcpyDeviceToHost));
      PT_HIP_CHECK(hipMemcpy(d_dep_ + (int64_t)ndep * dw, depx.data(),
                             depx.size() * 8, hipMemcpyHostToDevice));
      ndep += (unsigned long long)(depx.size() / dw);
      PT_HIP_CHECK(hipMemcpy(&d_ctr_[1], &ndep, 8, hipMemcpyHostToDevice));
    }
    for (int32_t j : out_j) {
      // outside the mesh: keep resident here at the requested origin
      const int64_t g = h_frame_[j];
      const double q[3] = {origin[j * 3], origin[j * 3 + 1],
                           origin[j * 3 + 2]};
      const int32_t none = -1;
      PT_HIP_CHECK(hipMemcpy(d_pos_ + g * 3, q, 24, hipMemcpyHostToDevice));
      PT_HIP_CHECK(
          hipMemcpy(d_elem_ + g, &none, 4, hipMemcpyHostToDevice));
    }
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

  std::vector<int32_t> elem_ids_global() const override {
    eng_->synchronize();
    std::vector<int32_t> le(n_);
    std::vector<uint8_t> r(n_);
    PT_HIP_CHECK(hipMemcpy(le.data(), d_elem_, n_ * 4,
                           hipMemcpyDeviceToHost));
    PT_HIP_CHECK(hipMemcpy(r.data(), d_res_, n_, hipMemcpyDeviceToHost));
    std::vector<int32_t> out(n_, -1);
    for (int64_t g = 0; g < n_; ++g)
      if (r[g] && le[g] >= 0) out[g] = dec_.l2g32[le[g]];
    return out;
  }

  std::vector<uint8_t> escaped_mask() const override {
    eng_->synchronize();
    std::vector<uint8_t> e(n_);
    PT_HIP_CHECK(hipMemcpy(e.data(), d_esc_, n_, hipMemcpyDeviceToHost));
    return e;
  }

  void set_state(const double *pos, const int32_t *gelem,
                 const uint8_t *escaped, int64_t n) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    eng_->synchronize();
    // ownership claims under THIS decomposition (host-side: once per
    // restore/repartition, not a hot path)
    std::vector<uint8_t> res(n_, 0);
    std::vector<int32_t> lel(n_, -1);
    for (int64_t g = 0; g < n_; ++g) {
      const int32_t ge = gelem[g];
      if (ge >= 0) {
        if (dec_.owners[ge] == rank_) {
          res[g] = 1;
          lel[g] = dec_.g2l[ge];
          if (lel[g] < 0)
            throw std::runtime_error(
                "set_state: owned element missing from this submesh");
        }
      } else if (rank_ == 0) {
        res[g] = 1; // out-of-mesh particles live on rank 0
      }
    }
    PT_HIP_CHECK(hipMemcpy(d_pos_, pos, n_ * 24, hipMemcpyHostToDevice));
    PT_HIP_CHECK(
        hipMemcpy(d_elem_, lel.data(), n_ * 4, hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_res_, res.data(), n_, hipMemcpyHostToDevice));
    PT_HIP_CHECK(
        hipMemcpy(d_esc_, escaped, n_, hipMemcpyHostToDevice));
    n_frame_ = -1;
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
                           const double *dest_host, const double *w_host,
                           const uint16_t *grp_host,
                           const double *resp_host) {
    std::vector<int32_t> gids(ne);
    PT_HIP_CHECK(
        hipMemcpy(gids.data(), d_eject_, ne * 4, hipMemcpyDeviceToHost));
    const int dw = rec_w_ + 1;
    std::vector<double> depx;
    std::vector<int32_t> outside;
    for (int64_t i = 0; i < ne; ++i) {
      const int64_t g = gids[i];
      const Vec3 q{origin[g * 3], origin[g * 3 + 1], origin[g * 3 + 2]};
      bool lo = false;
      const int32_t ge = full_locate_ ? full_locate_(q, loc_tol_, &lo) : -1;
      if (lo) stats_.loose_localizations++;
      if (ge >= 0) {
        std::vector<double> e(dw, 0.0);
        e[0] = (double)g;
        e[1] = q.x;
        e[2] = q.y;
        e[3] = q.z;
        e[4] = (double)ge;
        e[5] = dest_host[g * 3];
        e[6] = dest_host[g * 3 + 1];
        e[7] = dest_host[g * 3 + 2];
        e[8] = w_host[g];
        e[9] = 0.0; // fresh walk: no resume state
        e[10] = -1.0;
        int at = 11;
        if (carry_grp_) e[at++] = grp_host ? (double)grp_host[g] : 0.0;
        for (int k = 0; k < nscores_; ++k)
          e[at + k] = resp_host ? resp_host[g * nscores_ + k] : 1.0;
        e[rec_w_] = (double)dec_.owners[ge];
        depx.insert(depx.end(), e.begin(), e.end());
      } else {
        outside.push_back((int32_t)g);
      }
    }
    if (!depx.empty()) {
      // append to the device dep list (capacity n_*(rec_w_+1) is plenty:
      // ejects are a subset of residents)
      unsigned long long ndep = 0;
      PT_HIP_CHECK(hipMemcpy(&ndep, &d_ctr_[1], 8, hipMemcpyDeviceToHost));
      const int64_t m = (int64_t)depx.size() / dw;
      PT_HIP_CHECK(hipMemcpy(d_dep_ + (int64_t)ndep * dw, depx.data(),
                             depx.size() * 8, hipMemcpyHostToDevice));
      ndep += (unsigned long long)m;
      PT_HIP_CHECK(hipMemcpy(&d_ctr_[1], &ndep, 8, hipMemcpyHostToDevice));
      // the particle left this rank
      std::vector<uint8_t> zero(1, 0);
      for (int64_t i = 0; i < m; ++i) {
        const int64_t g = (int64_t)depx[(int64_t)i * dw];
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
  int rec_w_ = 10;
  bool carry_grp_ = false;
  int32_t *d_frame_ = nullptr;   // coupled-host frame (resident gids)
  int64_t n_frame_ = -1;         // -1: no snapshot taken
  std::vector<int64_t> h_frame_; // host copy of the frame
  double *d_ldest_ = nullptr, *d_lw_ = nullptr, *d_lorig_ = nullptr,
         *d_lresp_ = nullptr;
  int8_t *d_lfly_ = nullptr;
  uint16_t *d_lgrp_ = nullptr;
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
  // bitwise handoff-resume state (dep layout comment at the top)
  double *d_t0_ = nullptr, *d_wt0_ = nullptr;
  int32_t *d_prev_ = nullptr, *d_wprev_ = nullptr;
  double *d_wout_o_ = nullptr, *d_wout_t_ = nullptr;
  int32_t *d_wout_prev_ = nullptr;
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
    carry_grp_ = ngroups_ > 1;
    rec_w_ = 11 + (carry_grp_ ? 1 : 0) + nscores_;
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
      // strict hits only -- loose submesh hits can steal points whose
      // true element is absent from the ghost ring (GPU k_part_localize
      // comment); the full-mesh pass below resolves them
      if (le >= 0 && !lo && dec_.lowner[le] == rank_) {
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
    if (full_locate_) {
      for (int64_t g = 0; g < n_; ++g) {
        if ((claim[g >> 6] >> (g & 63)) & 1) continue;
        const Vec3 q{origins[g * 3], origins[g * 3 + 1], origins[g * 3 + 2]};
        bool lo = false;
        const int32_t ge = full_locate_(q, loc_tol_, &lo);
        if (ge < 0) continue;
        if (lo) stats_.loose_localizations++;
        claim[g >> 6] |= (int64_t)(1ull << (g & 63));
        if (dec_.owners[ge] == rank_) {
          res_[g] = 1;
          esc_[g] = 0;
          elem_[g] = dec_.g2l[ge];
          pos_[g * 3] = q.x;
          pos_[g * 3 + 1] = q.y;
          pos_[g * 3 + 2] = q.z;
        }
      }
    }
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

  // A resolved unit of walk work: every input the particle needs travels
  // with it, so global-array steps, coupled-host steps and exchanged
  // arrivals all feed the same loop (the CPU mirror of the GPU wire
  // format).
  struct Item {
    int64_t gid;
    Vec3 pos, dest;
    double w;
    uint16_t grp;
    std::vector<double> resp; // nscores entries when used, else empty
    int32_t lelem;
    // bitwise handoff resume (walk.h walk_segment doc): pos carries the
    // wrap-segment origin for exchanged arrivals; t0/prev seed the walk
    double t0 = 0.0;
    int32_t prev = -1;
  };

  void step(const double *dest, const int8_t *flying, const double *weights,
            int64_t n, const double *origin, const uint16_t *groups,
            const double *responses) override {
    check_n(n);
    if (groups && ngroups_ <= 1)
      throw std::runtime_error("groups passed but ngroups == 1");
    std::vector<Item> items;
    std::vector<double> dep;
    for (int64_t g = 0; g < n_; ++g) {
      if (!res_[g] || !flying[g]) continue;
      enter_particle(g, origin ? origin + g * 3 : nullptr, dest + g * 3,
                     weights[g], groups ? groups[g] : (uint16_t)0,
                     responses ? responses + g * nscores_ : nullptr,
                     responses != nullptr, items, dep);
    }
    run_items(std::move(items), std::move(dep), responses != nullptr);
    stats_.moves++;
  }

  std::vector<int64_t> resident_list() override {
    frame_.clear();
    for (int64_t g = 0; g < n_; ++g)
      if (res_[g]) frame_.push_back(g);
    have_frame_ = true;
    return frame_;
  }

  void step_local(const double *dest, const int8_t *flying,
                  const double *weights, int64_t n_local,
                  const double *origin, const uint16_t *groups,
                  const double *responses) override {
    if (!have_frame_)
      throw std::runtime_error(
          "step_local: call resident_list() first (it defines the input "
          "order this call consumes)");
    if (n_local != (int64_t)frame_.size())
      throw std::runtime_error("step_local: n_local " +
                               std::to_string(n_local) +
                               " != resident_list size " +
                               std::to_string(frame_.size()));
    if (groups && ngroups_ <= 1)
      throw std::runtime_error("groups passed but ngroups == 1");
    std::vector<Item> items;
    std::vector<double> dep;
    for (int64_t j = 0; j < n_local; ++j) {
      const int64_t g = frame_[j];
      if (!res_[g] || !flying[j]) continue;
      enter_particle(g, origin ? origin + j * 3 : nullptr, dest + j * 3,
                     weights[j], groups ? groups[j] : (uint16_t)0,
                     responses ? responses + j * nscores_ : nullptr,
                     responses != nullptr, items, dep);
    }
    have_frame_ = false; // residency changes below
    run_items(std::move(items), std::move(dep), responses != nullptr);
    stats_.moves++;
  }

  std::vector<double> flux_global() override {
    const std::vector<double> local = eng_->flux();
    const int64_t nl = dec_.sub.local.nelems;
    const int64_t slabs = (int64_t)nscores_ * ngroups_;
    std::vector<double> out(slabs * nelems_global_, 0.0);
    for (int64_t sl = 0; sl < slabs; ++sl)
      for (int64_t t = 0; t < nl; ++t)
        out[sl * nelems_global_ + dec_.sub.elem_l2g[t]] +=
            local[sl * nl + t];
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

  std::vector<int32_t> elem_ids_global() const override {
    std::vector<int32_t> out(n_, -1);
    for (int64_t g = 0; g < n_; ++g)
      if (res_[g] && elem_[g] >= 0) out[g] = dec_.l2g32[elem_[g]];
    return out;
  }

  std::vector<uint8_t> escaped_mask() const override { return esc_; }

  void set_state(const double *pos, const int32_t *gelem,
                 const uint8_t *escaped, int64_t n) override {
    check_n(n);
    for (int64_t g = 0; g < n_; ++g) {
      res_[g] = 0;
      elem_[g] = -1;
      const int32_t ge = gelem[g];
      if (ge >= 0) {
        if (dec_.owners[ge] == rank_) {
          res_[g] = 1;
          elem_[g] = dec_.g2l[ge];
          if (elem_[g] < 0)
            throw std::runtime_error(
                "set_state: owned element missing from this submesh");
        }
      } else if (rank_ == 0) {
        res_[g] = 1;
      }
      pos_[g * 3] = pos[g * 3];
      pos_[g * 3 + 1] = pos[g * 3 + 1];
      pos_[g * 3 + 2] = pos[g * 3 + 2];
      esc_[g] = escaped[g];
    }
    have_frame_ = false;
  }

public:
  std::function<int32_t(Vec3, double, bool *)> full_locate_;
  int max_rounds_ = 64;

private:
  void check_n(int64_t n) const {
    if (n != n_) throw std::runtime_error("global particle count mismatch");
  }

  // Phase A + admission for one flying resident particle: relocation /
  // reroute / eject resolution, then either queue a walk Item or emit a
  // departure record.
  void enter_particle(int64_t g, const double *origin3, const double *dest3,
                      double w, uint16_t grp, const double *resp,
                      bool resp_used, std::vector<Item> &items,
                      std::vector<double> &dep) {
    const Mesh &lm = dec_.sub.local;
    if (origin3 && !esc_[g]) {
      const Vec3 q{origin3[0], origin3[1], origin3[2]};
      if (q.x != pos_[g * 3] || q.y != pos_[g * 3 + 1] ||
          q.z != pos_[g * 3 + 2]) {
        stats_.relocated++;
        bool lo = false;
        int32_t le = lm.locate(q, loc_tol_, &lo);
        if (lo) le = -1; // loose submesh hit: resolve on the full mesh
        int64_t tgid = -1;
        int towner = -1;
        if (le >= 0 && dec_.lowner[le] != rank_) {
          tgid = dec_.l2g32[le];
          towner = dec_.lowner[le];
        } else if (le < 0) {
          bool lo2 = false;
          const int32_t ge =
              full_locate_ ? full_locate_(q, loc_tol_, &lo2) : -1;
          if (lo2) stats_.loose_localizations++;
          if (ge >= 0) {
            tgid = ge;
            towner = dec_.owners[ge];
          } else {
            elem_[g] = -1;
            pos_[g * 3] = q.x;
            pos_[g * 3 + 1] = q.y;
            pos_[g * 3 + 2] = q.z;
            return;
          }
        }
        if (towner >= 0) {
          emit_dep(dep, g, q, tgid,
                   Vec3{dest3[0], dest3[1], dest3[2]}, w, grp, resp,
                   resp_used, towner);
          res_[g] = 0;
          return;
        }
        elem_[g] = le;
        pos_[g * 3] = q.x;
        pos_[g * 3 + 1] = q.y;
        pos_[g * 3 + 2] = q.z;
      }
    }
    if (elem_[g] < 0) return; // outside mesh
    Item it;
    it.gid = g;
    it.pos = Vec3{pos_[g * 3], pos_[g * 3 + 1], pos_[g * 3 + 2]};
    it.dest = Vec3{dest3[0], dest3[1], dest3[2]};
    it.w = w;
    it.grp = grp;
    if (resp_used) it.resp.assign(resp, resp + nscores_);
    it.lelem = elem_[g];
    items.push_back(std::move(it));
  }

  void emit_dep(std::vector<double> &dep, int64_t g, Vec3 p, int64_t tgid,
                Vec3 d, double w, uint16_t grp, const double *resp,
                bool resp_used, int owner, double t = 0.0,
                int64_t prev_gid = -1) {
    const size_t base = dep.size();
    dep.resize(base + rec_w_ + 1, 0.0);
    double *e = dep.data() + base;
    e[0] = (double)g;
    e[1] = p.x;
    e[2] = p.y;
    e[3] = p.z;
    e[4] = (double)tgid;
    e[5] = d.x;
    e[6] = d.y;
    e[7] = d.z;
    e[8] = w;
    e[9] = t;
    e[10] = (double)prev_gid;
    int at = 11;
    if (carry_grp_) e[at++] = (double)grp;
    for (int k = 0; k < nscores_; ++k)
      e[at + k] = (resp_used && resp) ? resp[k] : 1.0;
    e[rec_w_] = (double)owner;
  }

  void run_items(std::vector<Item> items, std::vector<double> dep,
                 bool resp_used) {
    const bool groups_used = carry_grp_;
    const int dw = rec_w_ + 1;
    for (int round = 0; round <= max_rounds_; ++round) {
      if (!items.empty()) {
        const int64_t m = (int64_t)items.size();
        std::vector<double> wpos(m * 3), wdest(m * 3), ww(m),
            wout_pos(m * 3), wout_dest(m * 3);
        std::vector<int32_t> welem(m), wout_elem(m);
        std::vector<int8_t> wstatus(m);
        std::vector<uint16_t> wgrp(groups_used ? m : 0);
        std::vector<double> wresp(resp_used ? m * nscores_ : 0);
        std::vector<double> wt0(m), wout_o(m * 3), wout_t(m);
        std::vector<int32_t> wprev(m), wout_prev(m);
        for (int64_t j = 0; j < m; ++j) {
          const Item &it = items[j];
          wpos[j * 3] = it.pos.x;
          wpos[j * 3 + 1] = it.pos.y;
          wpos[j * 3 + 2] = it.pos.z;
          wdest[j * 3] = it.dest.x;
          wdest[j * 3 + 1] = it.dest.y;
          wdest[j * 3 + 2] = it.dest.z;
          welem[j] = it.lelem;
          ww[j] = it.w;
          wt0[j] = it.t0;
          wprev[j] = it.prev;
          if (groups_used) wgrp[j] = it.grp;
          if (resp_used)
            for (int k = 0; k < nscores_; ++k)
              wresp[j * nscores_ + k] = it.resp[k];
        }
        eng_->walk_raw(m, wpos.data(), wdest.data(), welem.data(), ww.data(),
                       wout_pos.data(), wout_elem.data(), wstatus.data(),
                       groups_used ? wgrp.data() : nullptr,
                       resp_used ? wresp.data() : nullptr,
                       wout_dest.data(), wt0.data(), wprev.data(),
                       wout_o.data(), wout_t.data(), wout_prev.data());
        for (int64_t j = 0; j < m; ++j) {
          const int64_t g = items[j].gid;
          if (wstatus[j] == 2) {
            const int32_t k = -(wout_elem[j] + 2);
            emit_dep(dep, g,
                     Vec3{wout_o[j * 3], wout_o[j * 3 + 1],
                          wout_o[j * 3 + 2]},
                     dec_.sub.foreign_gid[k],
                     Vec3{wout_dest[j * 3], wout_dest[j * 3 + 1],
                          wout_dest[j * 3 + 2]},
                     ww[j], groups_used ? wgrp[j] : (uint16_t)0,
                     resp_used ? &wresp[j * nscores_] : nullptr, resp_used,
                     dec_.sub.foreign_owner[k], wout_t[j],
                     wout_prev[j] >= 0 ? (int64_t)dec_.l2g32[wout_prev[j]]
                                       : (int64_t)-1);
            res_[g] = 0;
          } else {
            pos_[g * 3] = wout_pos[j * 3];
            pos_[g * 3 + 1] = wout_pos[j * 3 + 1];
            pos_[g * 3 + 2] = wout_pos[j * 3 + 2];
            elem_[g] = wout_elem[j];
            esc_[g] = (wstatus[j] == 1) ? 1 : 0;
          }
        }
      }
      items.clear();

      // bucket by owner, exchange, unpack into next-round items
      const int64_t m = (int64_t)dep.size() / dw;
      std::vector<int64_t> scounts(world_, 0);
      std::vector<double> send(m * rec_w_);
      {
        std::vector<int64_t> offs(world_, 0), cur(world_, 0);
        for (int64_t i = 0; i < m; ++i)
          scounts[(int)dep[i * dw + rec_w_]]++;
        int64_t acc = 0;
        for (int r = 0; r < world_; ++r) {
          offs[r] = acc;
          acc += scounts[r];
        }
        for (int64_t i = 0; i < m; ++i) {
          const int o = (int)dep[i * dw + rec_w_];
          const int64_t sl = offs[o] + cur[o]++;
          for (int k = 0; k < rec_w_; ++k)
            send[sl * rec_w_ + k] = dep[i * dw + k];
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
        for (int r = 0; r < world_; ++r) sc[r] = scounts[r] * rec_w_;
        recv = comm_->alltoallv(send.data(), sc);
      }
      const int64_t nr = (int64_t)recv.size() / rec_w_;
      for (int64_t i = 0; i < nr; ++i) {
        const double *e = recv.data() + i * rec_w_;
        const int64_t g = (int64_t)e[0];
        res_[g] = 1;
        esc_[g] = 0;
        pos_[g * 3] = e[1];
        pos_[g * 3 + 1] = e[2];
        pos_[g * 3 + 2] = e[3];
        elem_[g] = dec_.g2l[(int64_t)e[4]];
        Item it;
        it.gid = g;
        it.pos = Vec3{e[1], e[2], e[3]};
        it.dest = Vec3{e[5], e[6], e[7]};
        it.w = e[8];
        it.t0 = e[9];
        const int64_t pg = (int64_t)e[10];
        it.prev = pg >= 0 ? dec_.g2l[pg] : -1;
        int at = 11;
        it.grp = carry_grp_ ? (uint16_t)e[at++] : (uint16_t)0;
        if (resp_used) it.resp.assign(e + at, e + at + nscores_);
        it.lelem = elem_[g];
        items.push_back(std::move(it));
      }
      if (round == max_rounds_)
        throw std::runtime_error("partitioned step did not converge in " +
                                 std::to_string(max_rounds_) +
                                 " handoff rounds");
    }
  }

  int64_t n_;
  Comm *comm_;
  int rank_, world_;
  int ngroups_, nscores_ = 1;
  int rec_w_ = 10;
  bool carry_grp_ = false;
  int64_t nelems_global_;
  Decomp dec_;
  std::unique_ptr<Engine> eng_;
  double loc_tol_ = 0.0;
  mutable EngineStats stats_;
  std::vector<double> pos_;
  std::vector<int32_t> elem_;
  std::vector<uint8_t> res_, esc_;
  std::vector<int64_t> frame_;
  bool have_frame_ = false;
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
