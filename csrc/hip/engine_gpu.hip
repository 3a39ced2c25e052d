// MI355X (gfx950) tally engine.
//
// Design (MI355X-first, not a port of the reference's Kokkos structure):
//   * The whole per-step pipeline of the reference -- H2D staging (M1-M3),
//     buffer->particle copies (K1-K4), the external ParticleTracer search
//     loop and the per-boundary handler kernels (K5-K9)
//     (/root/reference/src/pumitally/PumiTallyImpl.cpp:66-149,243-380) --
//     collapses into ONE fused kernel (k_move) that walks each particle's
//     whole segment in registers and atomicAdds per-element contributions.
//   * Mesh data is flat HBM arrays consumed by the walk: 4 face planes
//     (double4, inward-positive unit normals) + 4 neighbor ids per tet --
//     144 B/tet, contiguous, no indirection to vertex coords in the hot
//     loop.  A 1M-tet mesh (~150 MB) is resident in the 256 MiB
//     Infinity Cache.
//   * Particle state lives in SPATIAL (Morton) order: a device radix sort
//     (hipCUB) orders particle slots by position at localization and
//     periodically thereafter, so neighboring lanes walk neighboring
//     tets and each XCD's private L2 sees one compact mesh region
//     (measured 2.2x on the walk).  A slot->caller index map keeps the
//     public API's particle indices unchanged; the caller's arrays are
//     gathered through it on device.
//   * Host input staging is parity double-buffered: step k+1's H2D
//     copies (origin/dest/flying/weights, plus groups/responses when
//     used; on the copy stream) overlap step k's walk kernels (on the
//     compute stream); buffer reuse is fenced with per-parity events.
//     Pinned sources (pumiumtally_amd.pinned_array or app-registered
//     buffers) run at full link rate.
#include "../core/engine.h"
#include "../core/walk.h"

#include <hip/hip_runtime.h>
#include <hipcub/hipcub.hpp>

#include <algorithm>
#include <type_traits>
#include <array>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

namespace pumitally {

#define PT_HIP_CHECK(expr)                                                     \
  do {                                                                         \
    hipError_t _e = (expr);                                                    \
    if (_e != hipSuccess)                                                      \
      throw std::runtime_error(std::string("HIP error at " __FILE__ ":") +     \
                               std::to_string(__LINE__) + ": " +               \
                               hipGetErrorString(_e));                         \
  } while (0)

namespace {

// Block size for all particle kernels; swept via PUMITALLY_BLOCK.
inline int block_size() {
  static int v = [] {
    const char *s = getenv("PUMITALLY_BLOCK");
    int k = s ? atoi(s) : 256;
    if (k < 64) k = 64;
    if (k > 1024) k = 1024;
    return (k / 64) * 64; // multiple of the 64-wide wavefront
  }();
  return v;
}
#define kBlock block_size()

__global__ void k_init_particles(double *__restrict__ pos,
                                 int32_t *__restrict__ elem,
                                 uint8_t *__restrict__ escaped,
                                 int32_t *__restrict__ s2c, int64_t n,
                                 double cx, double cy, double cz) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    pos[i * 3] = cx;
    pos[i * 3 + 1] = cy;
    pos[i * 3 + 2] = cz;
    elem[i] = 0;
    escaped[i] = 0;
    s2c[i] = (int32_t)i;
  }
}

__global__ void k_iota(int32_t *__restrict__ a, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride)
    a[i] = (int32_t)i;
}

__global__ void k_locate(const Plane *__restrict__ planes, GridView grid,
                         const double *__restrict__ q,
                         double *__restrict__ pos, int32_t *__restrict__ elem,
                         uint8_t *__restrict__ escaped,
                         unsigned long long *__restrict__ loose, int64_t n,
                         double tol) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const Vec3 p{q[i * 3], q[i * 3 + 1], q[i * 3 + 2]};
    bool used_loose = false;
    elem[i] = grid_locate(grid, planes, p, tol, &used_loose);
    if (used_loose) atomicAdd(loose, 1ull); // rare by construction
    pos[i * 3] = p.x;
    pos[i * 3 + 1] = p.y;
    pos[i * 3 + 2] = p.z;
    escaped[i] = 0;
  }
}

__device__ __forceinline__ uint64_t morton_spread(uint64_t v) {
  v &= 0x1fffff;
  v = (v | v << 32) & 0x1f00000000ffffull;
  v = (v | v << 16) & 0x1f0000ff0000ffull;
  v = (v | v << 8) & 0x100f00f00f00f00full;
  v = (v | v << 4) & 0x10c30c30c30c30c3ull;
  v = (v | v << 2) & 0x1249249249249249ull;
  return v;
}

__global__ void k_morton_keys(const double *__restrict__ pos,
                              uint64_t *__restrict__ keys,
                              int32_t *__restrict__ vals, int64_t n, Vec3 lo,
                              Vec3 scale) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    double x = (pos[i * 3] - lo.x) * scale.x;
    double y = (pos[i * 3 + 1] - lo.y) * scale.y;
    double z = (pos[i * 3 + 2] - lo.z) * scale.z;
    x = x < 0 ? 0 : (x > 2097151.0 ? 2097151.0 : x);
    y = y < 0 ? 0 : (y > 2097151.0 ? 2097151.0 : y);
    z = z < 0 ? 0 : (z > 2097151.0 ? 2097151.0 : z);
    keys[i] = morton_spread((uint64_t)x) | (morton_spread((uint64_t)y) << 1) |
              (morton_spread((uint64_t)z) << 2);
    vals[i] = (int32_t)i;
  }
}

__global__ void k_permute_state(const int32_t *__restrict__ order,
                                const double *__restrict__ pos_in,
                                const int32_t *__restrict__ elem_in,
                                const uint8_t *__restrict__ esc_in,
                                const int32_t *__restrict__ s2c_in,
                                double *__restrict__ pos_out,
                                int32_t *__restrict__ elem_out,
                                uint8_t *__restrict__ esc_out,
                                int32_t *__restrict__ s2c_out, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t j = blockIdx.x * blockDim.x + threadIdx.x; j < n; j += stride) {
    const int32_t i = order[j];
    pos_out[j * 3] = pos_in[(int64_t)i * 3];
    pos_out[j * 3 + 1] = pos_in[(int64_t)i * 3 + 1];
    pos_out[j * 3 + 2] = pos_in[(int64_t)i * 3 + 2];
    elem_out[j] = elem_in[i];
    esc_out[j] = esc_in[i];
    s2c_out[j] = s2c_in[i];
  }
}

__global__ void k_accumulate_batch(double *__restrict__ flux,
                                   double *__restrict__ bsum,
                                   double *__restrict__ bsq, int64_t n) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < n; i += stride) {
    const double f = flux[i];
    bsum[i] += f;
    bsq[i] += f * f;
    flux[i] = 0.0;
  }
}

__global__ void k_reduce_slices(double *__restrict__ flux, int64_t nelems,
                                int slices) {
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t i = blockIdx.x * blockDim.x + threadIdx.x; i < nelems;
       i += stride) {
    double s = flux[i];
    for (int k = 1; k < slices; ++k) {
      s += flux[(int64_t)k * nelems + i];
      flux[(int64_t)k * nelems + i] = 0.0;
    }
    flux[i] = s;
  }
}

// Wave-level tally aggregation: Morton-sorted slots put neighboring lanes
// in neighboring tets, so at a given walk iteration many of a wave's 64
// lanes tally into the SAME element.  The fp64-atomic pipe is the measured
// ceiling of the walk (profiles/README.md: chase probe 19.1 G atomics/s;
// walk at 84% of it), while VALU sits ~12% busy -- so spend idle VALU on a
// segmented wave reduction over runs of equal element id and issue ONE
// atomicAdd per run (the run tail carries the total).  Runs are maximal
// CONTIGUOUS ACTIVE lane spans with equal el: a gap in the active mask
// starts a new run, so shuffles never read across inactive lanes.
__device__ __forceinline__ void wave_agg_atomic_add(double *__restrict__ flux,
                                                    int32_t el, double v) {
  const unsigned long long mask = __ballot(1);
  const int lane = (int)(threadIdx.x & 63u);
  const int32_t prev_el = __shfl_up(el, 1);
  const bool prev_active = lane > 0 && ((mask >> (lane - 1)) & 1ull);
  const bool head = !prev_active || prev_el != el;
  // inclusive max-scan of head lane indices: every lane learns its run head
  int head_lane = head ? lane : 0;
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) {
    const int up = __shfl_up(head_lane, off);
    if (lane >= off && ((mask >> (lane - off)) & 1ull) && up > head_lane)
      head_lane = up;
  }
  // inclusive segmented sum over the run (all source lanes inside the run
  // are active and contiguous by construction)
  double acc = v;
#pragma unroll
  for (int off = 1; off < 64; off <<= 1) {
    const double up = __shfl_up(acc, off);
    if (lane - off >= head_lane) acc += up;
  }
  const int32_t next_el = __shfl_down(el, 1);
  const bool next_active = lane < 63 && ((mask >> (lane + 1)) & 1ull);
  if (!next_active || next_el != el) atomicAdd(&flux[el], acc);
}

inline bool wave_agg() {
  static bool v = [] {
    const char *s = getenv("PUMITALLY_WAVE_AGG");
    // measured on MI355X: device-resident walk 1776 -> 1831M ps/s
    // (chord 8, sorted slots), flux bit-identical; default on
    return s ? atoi(s) != 0 : true;
  }();
  return v;
}

// The fused move kernel: phase A (relocation of flying, non-escaped
// particles whose origin changed) + phase B (tallied walk to destination).
//
// Iteration is over particle SLOTS (spatial order); the caller's arrays
// are gathered through s2c.  Block->slot mapping is XCD-aware: MI355X
// dispatches block b to XCD b%8 and each XCD has a private 4 MiB L2; we
// remap blocks so each XCD owns one contiguous (hence spatially compact)
// slot range.  Purely a speed lever: any placement is correct.
template <bool F32, bool Scored = false, bool Periodic = false,
          bool Agg = false>
__global__ void k_move(const Plane *__restrict__ planes,
                       const Plane32 *__restrict__ planes32,
                       const int32_t *__restrict__ nbr, GridView grid,
                       const int32_t *__restrict__ s2c,
                       const double *__restrict__ origin,
                       const double *__restrict__ dest,
                       const int8_t *__restrict__ flying,
                       const double *__restrict__ weights,
                       const uint16_t *__restrict__ groups, int ngroups,
                       double *__restrict__ pos, int32_t *__restrict__ elem,
                       uint8_t *__restrict__ escaped,
                       double *__restrict__ flux,
                       unsigned long long *__restrict__ lost,
                       double *__restrict__ lostrec,
                       unsigned long long *__restrict__ loose, int64_t lo,
                       int64_t hi, double loc_tol, int max_steps,
                       int64_t nelems, int slice_mask, bool reflective,
                       const uint32_t *__restrict__ face_bc,
                       const double *__restrict__ resp = nullptr,
                       int nscores = 1,
                       const int32_t *__restrict__ pidx = nullptr,
                       const int32_t *__restrict__ pelem = nullptr,
                       const double *__restrict__ pshift = nullptr) {
  // Scored=false, Periodic=false is the headline instantiation:
  // resp/nscores/pidx are unused, the slice stride stays nelems*ngroups,
  // and the walk compiles without the score loop or the periodic branch --
  // codegen identical to before those features existed.
  flux += (int64_t)(blockIdx.x & (unsigned)slice_mask) * nelems * ngroups *
          (Scored ? nscores : 1);
  const unsigned bpx = gridDim.x / 8u;
  const unsigned vb = (blockIdx.x % 8u) * bpx + blockIdx.x / 8u;
  const int64_t m = hi - lo;
  const int64_t per_blk = (m + gridDim.x - 1) / gridDim.x;
  const int64_t base = lo + (int64_t)vb * per_blk;
  const int64_t end = base + per_blk < hi ? base + per_blk : hi;

  for (int64_t i = base + threadIdx.x; i < end; i += blockDim.x) {
    const int64_t c = s2c[i];
    if (!flying[c]) continue;
    Vec3 o{pos[i * 3], pos[i * 3 + 1], pos[i * 3 + 2]};
    int32_t e = elem[i];
    if (origin != nullptr && !escaped[i]) {
      const Vec3 q{origin[c * 3], origin[c * 3 + 1], origin[c * 3 + 2]};
      if (q.x != o.x || q.y != o.y || q.z != o.z) {
        bool used_loose = false;
        e = grid_locate(grid, planes, q, loc_tol, &used_loose);
        if (used_loose) atomicAdd(loose, 1ull);
        o = q;
      }
    }
    if (e < 0) {
      pos[i * 3] = o.x;
      pos[i * 3 + 1] = o.y;
      pos[i * 3 + 2] = o.z;
      elem[i] = e;
      continue;
    }
    const Vec3 d{dest[c * 3], dest[c * 3 + 1], dest[c * 3 + 2]};
    int32_t out_elem;
    Vec3 out_pos;
    bool out_esc;
    const int64_t goff =
        groups ? (int64_t)(groups[c] % ngroups) * nelems : 0;
    auto add = [&](int32_t el, double v) {
      if constexpr (Scored) {
        const int64_t gsz = (int64_t)ngroups * nelems;
        if (resp) {
          for (int k = 0; k < nscores; ++k)
            atomicAdd(&flux[k * gsz + goff + el], v * resp[c * nscores + k]);
        } else {
          atomicAdd(&flux[goff + el], v);
        }
      } else if constexpr (Agg) {
        // dispatched only when groups==nullptr, so goff==0 for every lane
        // and el alone is the aggregation key
        wave_agg_atomic_add(flux, el, v);
      } else {
        atomicAdd(&flux[goff + el], v);
      }
    };
    if constexpr (F32)
      walk_segment32<Periodic>(planes, planes32, nbr, e, o, d, weights[c],
                               max_steps, add, &out_elem, &out_pos, &out_esc,
                               reflective, face_bc, pidx, pelem, pshift);
    else
      walk_segment<Periodic>(planes, nbr, e, o, d, weights[c], max_steps,
                             add, &out_elem, &out_pos, &out_esc, reflective,
                             face_bc, pidx, pelem, pshift);
    if (out_elem == kWalkLost) {
      const unsigned long long k = atomicAdd(lost, 1ull);
      if (k < (unsigned long long)kMaxLostRecords) {
        lostrec[k * 4] = (double)c;
        lostrec[k * 4 + 1] = out_pos.x;
        lostrec[k * 4 + 2] = out_pos.y;
        lostrec[k * 4 + 3] = out_pos.z;
      }
      out_elem = e;
    }
    elem[i] = out_elem;
    pos[i * 3] = out_pos.x;
    pos[i * 3 + 1] = out_pos.y;
    pos[i * 3 + 2] = out_pos.z;
    escaped[i] = out_esc ? 1 : 0;
  }
}

template <bool F32>
__global__ void k_walk_raw(const Plane *__restrict__ planes,
                           const Plane32 *__restrict__ planes32,
                           const int32_t *__restrict__ nbr,
                           const double *__restrict__ pos,
                           const double *__restrict__ dest,
                           const int32_t *__restrict__ elem,
                           const double *__restrict__ weights,
                           const uint16_t *__restrict__ groups,
                           const double *__restrict__ resp,
                           double *__restrict__ out_pos,
                           int32_t *__restrict__ out_elem,
                           int8_t *__restrict__ out_status,
                           double *__restrict__ out_dest,
                           double *__restrict__ flux,
                           unsigned long long *__restrict__ lost,
                           double *__restrict__ lostrec, int64_t n,
                           int max_steps, bool reflective,
                           const uint32_t *__restrict__ face_bc, int ngroups,
                           int64_t nelems, int nscores,
                           const int32_t *__restrict__ pidx = nullptr,
                           const int32_t *__restrict__ pelem = nullptr,
                           const double *__restrict__ pshift = nullptr,
                           const double *__restrict__ in_t = nullptr,
                           const int32_t *__restrict__ in_prev = nullptr,
                           double *__restrict__ out_o = nullptr,
                           double *__restrict__ out_t = nullptr,
                           int32_t *__restrict__ out_prev = nullptr) {
  // XCD-aware block->range remap, same as k_move: MI355X dispatches
  // block b to XCD b%8; giving each XCD one contiguous (spatially
  // compact, since callers keep lists near-Morton-ordered) index range
  // keeps its private L2 on one mesh region.
  const unsigned bpx = gridDim.x / 8u;
  const unsigned vb = bpx ? (blockIdx.x % 8u) * bpx + blockIdx.x / 8u
                          : blockIdx.x;
  const int64_t per_blk = (n + gridDim.x - 1) / gridDim.x;
  const int64_t base = (int64_t)vb * per_blk;
  const int64_t end = base + per_blk < n ? base + per_blk : n;
  for (int64_t i = base + threadIdx.x; i < end; i += blockDim.x) {
    const Vec3 o{pos[i * 3], pos[i * 3 + 1], pos[i * 3 + 2]};
    const Vec3 d{dest[i * 3], dest[i * 3 + 1], dest[i * 3 + 2]};
    int32_t oe;
    Vec3 op;
    bool esc;
    const int64_t goff = groups ? (int64_t)(groups[i] % ngroups) * nelems : 0;
    const int64_t gsz = (int64_t)ngroups * nelems;
    auto add = [&](int32_t e, double v) {
      if (resp) {
        for (int k = 0; k < nscores; ++k)
          atomicAdd(&flux[k * gsz + goff + e], v * resp[i * nscores + k]);
      } else {
        atomicAdd(&flux[goff + e], v);
      }
    };
    Vec3 od{d.x, d.y, d.z};
    const double rt = in_t ? in_t[i] : 0.0;
    const int32_t rp = in_prev ? in_prev[i] : -1;
    Vec3 oo{o.x, o.y, o.z};
    double ot = 0.0;
    int32_t opv = -1;
    if constexpr (F32)
      walk_segment32<true>(planes, planes32, nbr, elem[i], o, d, weights[i],
                           max_steps, add, &oe, &op, &esc, reflective,
                           face_bc, pidx, pelem, pshift, &od, rt, rp, &oo,
                           &ot, &opv);
    else
      walk_segment<true>(planes, nbr, elem[i], o, d, weights[i], max_steps,
                         add, &oe, &op, &esc, reflective, face_bc, pidx,
                         pelem, pshift, &od, rt, rp, &oo, &ot, &opv);
    int8_t st = 0;
    if (oe == kWalkLost) {
      st = 3;
      oe = elem[i];
      const unsigned long long k = atomicAdd(lost, 1ull);
      if (k < (unsigned long long)kMaxLostRecords) {
        lostrec[k * 4] = (double)i;
        lostrec[k * 4 + 1] = op.x;
        lostrec[k * 4 + 2] = op.y;
        lostrec[k * 4 + 3] = op.z;
      }
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

// Tally slices: S independent copies of the flux array, selected by
// blockIdx & (S-1), reduced at readout.  The hot-element (point-source)
// mitigation: measured on MI355X (profiles/README.md round 2), the
// config-4 contention stress at 10M particles through a 2%-cube source
// runs 74.2 ms/step with 1 slice, 7.8 ms with 64 and 6.48 ms with 256
// (11.4x, within 18% of the uncontended walk), while spread sources and
// the headline are unchanged to the noise floor at any slice count.
// Default is adaptive: 256 slices when the copies fit a 2 GiB HBM
// budget (trivial against 288 GB for 1M-tet meshes), halved until they
// do.
int flux_slices(int64_t flux_doubles) {
  const char *s = getenv("PUMITALLY_FLUX_SLICES");
  int k = s ? atoi(s) : 256;
  if (k < 1) k = 1;
  if (k > 256) k = 256;
  while (k & (k - 1)) k--; // power of two for the cheap in-kernel mask
  if (!s) {
    // adaptive: keep the slice copies under ~2 GiB
    const int64_t budget = (int64_t)2 << 30;
    while (k > 1 && flux_doubles * k * 8 > budget) k >>= 1;
  }
  return k;
}

int grid_cap() {
  static int cap = [] {
    const char *s = getenv("PUMITALLY_GRID_CAP");
    return s ? atoi(s) : 1024; // swept on MI355X: 1024 > 2048 > 4096
  }();
  return cap;
}

int sort_every() {
  static int v = [] {
    const char *s = getenv("PUMITALLY_SORT");
    return s ? atoi(s) : 16; // re-sort cadence in moves; 0 disables
  }();
  return v;
}

int64_t chunk_particles(int64_t n) {
  static int64_t c = [] {
    const char *s = getenv("PUMITALLY_CHUNK");
    return s ? (int64_t)atoll(s) : (int64_t)0;
  }();
  if (c > 0) return c;
  // swept on MI355X: ~2.6M-particle chunks beat 1.25M and whole-batch
  return std::max<int64_t>((int64_t)2621440, (n + 7) / 8);
}

int grid_blocks(int64_t work) {
  int64_t blocks = (work + kBlock - 1) / kBlock;
  if (blocks > grid_cap()) blocks = grid_cap();
  // round up to a multiple of 8 for the XCD-aware remap in k_move
  return (int)((blocks + 7) / 8 * 8);
}

template <class T> T *dmalloc(int64_t count) {
  void *p = nullptr;
  PT_HIP_CHECK(hipMalloc(&p, count * sizeof(T)));
  return (T *)p;
}

class GpuEngine final : public Engine {
public:
  GpuEngine(Mesh mesh, int64_t n, int device, int groups, int scores)
      : mesh_(std::move(mesh)), n_(n) {
    ngroups = groups < 1 ? 1 : groups;
    nscores = scores < 1 ? 1 : scores;
    PT_HIP_CHECK(hipSetDevice(device));
    device_ = device;
    PT_HIP_CHECK(hipStreamCreateWithFlags(&s_copy_, hipStreamNonBlocking));
    PT_HIP_CHECK(hipStreamCreateWithFlags(&s_comp_, hipStreamNonBlocking));
    for (auto &ev : copy_ev_)
      PT_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));
    for (auto &ev : kernels_done_)
      PT_HIP_CHECK(hipEventCreateWithFlags(&ev, hipEventDisableTiming));

    // Mesh upload (once).
    d_planes_ = dmalloc<Plane>(mesh_.nelems * 4);
    d_planes32_ = dmalloc<Plane32>(mesh_.nelems * 4);
    d_nbr_ = dmalloc<int32_t>(mesh_.nelems * 4);
    d_cell_start_ = dmalloc<int32_t>(mesh_.grid.cell_start.size());
    d_cell_tets_ = dmalloc<int32_t>(mesh_.grid.cell_tets.size() + 1);
    PT_HIP_CHECK(hipMemcpy(d_planes_, mesh_.planes.data(),
                           mesh_.nelems * 4 * sizeof(Plane), hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_planes32_, mesh_.planes32.data(),
                           mesh_.nelems * 4 * sizeof(Plane32), hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_nbr_, mesh_.nbr.data(),
                           mesh_.nelems * 4 * sizeof(int32_t), hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_cell_start_, mesh_.grid.cell_start.data(),
                           mesh_.grid.cell_start.size() * sizeof(int32_t),
                           hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_cell_tets_, mesh_.grid.cell_tets.data(),
                           mesh_.grid.cell_tets.size() * sizeof(int32_t),
                           hipMemcpyHostToDevice));
    if (!mesh_.face_bc_bits.empty()) {
      d_face_bc_ = dmalloc<uint32_t>(mesh_.face_bc_bits.size());
      PT_HIP_CHECK(hipMemcpy(d_face_bc_, mesh_.face_bc_bits.data(),
                             mesh_.face_bc_bits.size() * sizeof(uint32_t),
                             hipMemcpyHostToDevice));
    }
    if (mesh_.has_periodic()) {
      d_pidx_ = dmalloc<int32_t>(mesh_.periodic_idx.size());
      d_pelem_ = dmalloc<int32_t>(mesh_.periodic_elem.size());
      d_pshift_ = dmalloc<double>(mesh_.periodic_shift.size());
      PT_HIP_CHECK(hipMemcpy(d_pidx_, mesh_.periodic_idx.data(),
                             mesh_.periodic_idx.size() * 4,
                             hipMemcpyHostToDevice));
      PT_HIP_CHECK(hipMemcpy(d_pelem_, mesh_.periodic_elem.data(),
                             mesh_.periodic_elem.size() * 4,
                             hipMemcpyHostToDevice));
      PT_HIP_CHECK(hipMemcpy(d_pshift_, mesh_.periodic_shift.data(),
                             mesh_.periodic_shift.size() * 8,
                             hipMemcpyHostToDevice));
    }
    grid_view_ = GridView{mesh_.grid.nx, mesh_.grid.ny, mesh_.grid.nz,
                          mesh_.grid.lo,  mesh_.grid.inv_h,
                          d_cell_start_,  d_cell_tets_};

    // Particle state (slot order) + slot->caller map.
    d_pos_ = dmalloc<double>(n_ * 3);
    d_elem_ = dmalloc<int32_t>(n_);
    d_escaped_ = dmalloc<uint8_t>(n_);
    d_s2c_ = dmalloc<int32_t>(n_);
    fsz_ = mesh_.nelems * ngroups * nscores;
    slices_ = flux_slices(fsz_);
    d_flux_ = dmalloc<double>(fsz_ * slices_);
    d_lost_ = dmalloc<unsigned long long>(1);
    d_loose_ = dmalloc<unsigned long long>(1);
    d_lostrec_ = dmalloc<double>((int64_t)kMaxLostRecords * 4);
    PT_HIP_CHECK(hipMemset(d_flux_, 0, fsz_ * slices_ * sizeof(double)));
    PT_HIP_CHECK(hipMemset(d_lost_, 0, sizeof(unsigned long long)));
    PT_HIP_CHECK(hipMemset(d_loose_, 0, sizeof(unsigned long long)));
    PT_HIP_CHECK(
        hipMemset(d_lostrec_, 0, kMaxLostRecords * 4 * sizeof(double)));

    // Parity double-buffered input staging.
    for (int p = 0; p < 2; ++p) {
      d_origin_[p] = dmalloc<double>(n_ * 3);
      d_dest_[p] = dmalloc<double>(n_ * 3);
      d_flying_[p] = dmalloc<int8_t>(n_);
      d_weights_[p] = dmalloc<double>(n_);
      d_groups_[p] = ngroups > 1 ? dmalloc<uint16_t>(n_) : nullptr;
      // d_resp_ is lazy-allocated on the first move() that passes
      // responses (valid even with nscores==1: a single per-particle
      // response multiplier).
    }

    // Spatial-sort scratch.
    sort_every_ = sort_every();
    if (sort_every_ > 0) {
      d_keys_ = dmalloc<uint64_t>(n_);
      d_keys2_ = dmalloc<uint64_t>(n_);
      d_vals_ = dmalloc<int32_t>(n_);
      d_order_ = dmalloc<int32_t>(n_);
      d_pos2_ = dmalloc<double>(n_ * 3);
      d_elem2_ = dmalloc<int32_t>(n_);
      d_esc2_ = dmalloc<uint8_t>(n_);
      d_s2c2_ = dmalloc<int32_t>(n_);
      size_t bytes = 0;
      PT_HIP_CHECK(hipcub::DeviceRadixSort::SortPairs(
          nullptr, bytes, d_keys_, d_keys2_, d_vals_, d_order_, (int)n_, 0, 64,
          s_comp_));
      sorttmp_bytes_ = bytes;
      PT_HIP_CHECK(hipMalloc(&d_sorttmp_, bytes ? bytes : 1));
      const Vec3 ext = mesh_.bbox_hi - mesh_.bbox_lo;
      sort_scale_ = Vec3{ext.x > 0 ? 2097151.0 / ext.x : 0.0,
                         ext.y > 0 ? 2097151.0 / ext.y : 0.0,
                         ext.z > 0 ? 2097151.0 / ext.z : 0.0};
    }

    loc_tol_ = loc_tol_rel() * norm(mesh_.bbox_hi - mesh_.bbox_lo);
    walk_fp32 = default_walk_fp32();
    reflective = default_reflective();
    const Vec3 c0 = mesh_.nelems > 0 ? mesh_.centroid(0) : Vec3{0, 0, 0};
    k_init_particles<<<grid_blocks(n_), kBlock, 0, s_comp_>>>(
        d_pos_, d_elem_, d_escaped_, d_s2c_, n_, c0.x, c0.y, c0.z);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
  }

  ~GpuEngine() override {
    (void)hipSetDevice(device_);
    (void)hipDeviceSynchronize();
    for (void *p : {(void *)d_planes_, (void *)d_planes32_, (void *)d_nbr_,
                    (void *)d_cell_start_, (void *)d_cell_tets_,
                    (void *)d_pos_, (void *)d_elem_, (void *)d_escaped_,
                    (void *)d_s2c_, (void *)d_flux_, (void *)d_lost_,
                    (void *)d_loose_, (void *)d_lostrec_,
                    (void *)d_origin_[0], (void *)d_origin_[1],
                    (void *)d_dest_[0], (void *)d_dest_[1],
                    (void *)d_flying_[0], (void *)d_flying_[1],
                    (void *)d_weights_[0], (void *)d_weights_[1],
                    (void *)d_groups_[0], (void *)d_groups_[1],
                    (void *)d_resp_[0], (void *)d_resp_[1],
                    (void *)d_bsum_, (void *)d_bsq_, (void *)d_face_bc_,
                    (void *)d_pidx_, (void *)d_pelem_, (void *)d_pshift_,
                    (void *)d_keys_, (void *)d_keys2_, (void *)d_vals_,
                    (void *)d_order_, (void *)d_pos2_, (void *)d_elem2_,
                    (void *)d_esc2_, (void *)d_s2c2_, d_sorttmp_})
      if (p) (void)hipFree(p);
    free_wr();
    for (auto &ev : copy_ev_) (void)hipEventDestroy(ev);
    for (auto &ev : kernels_done_) (void)hipEventDestroy(ev);
    (void)hipStreamDestroy(s_copy_);
    (void)hipStreamDestroy(s_comp_);
  }

  int64_t num_particles() const override { return n_; }
  const Mesh &mesh() const override { return mesh_; }

  void copy_initial_position(const double *p, int64_t n) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    sync(); // staging buffers and state may be read by in-flight kernels
    stage(p, n * 3 * sizeof(double), d_origin_[0], s_copy_);
    PT_HIP_CHECK(hipStreamSynchronize(s_copy_));
    k_iota<<<grid_blocks(n_), kBlock, 0, s_comp_>>>(d_s2c_, n_);
    k_locate<<<grid_blocks(n_), kBlock, 0, s_comp_>>>(
        d_planes_, grid_view_, d_origin_[0], d_pos_, d_elem_, d_escaped_,
        d_loose_, n_, loc_tol_);
    PT_HIP_CHECK(hipGetLastError());
    if (sort_every_ > 0) spatial_sort();
    moves_since_sort_ = 0;
    // k_locate reads the staging buffer d_origin_[0]; an immediately
    // following move() would re-stage into it with no recorded fence
    // (kernels_done_[0] has never been recorded at that point).  This is
    // a once-per-batch call: block until localization/sort finish.
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
  }

  void move(const double *origin, const double *dest, const int8_t *flying,
            const double *weights, int64_t n,
            const uint16_t *groups = nullptr,
            const double *responses = nullptr) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    const int steps = max_steps > 0 ? max_steps : default_max_steps(mesh_);
    const int p = (int)(parity_++ & 1);
    // WAR fence: buffer set p was last read by the kernels of the move
    // two calls ago; its copies must wait for them.
    PT_HIP_CHECK(hipStreamWaitEvent(s_copy_, kernels_done_[p], 0));
    if (origin)
      stage(origin, n * 3 * sizeof(double), d_origin_[p], s_copy_);
    stage(dest, n * 3 * sizeof(double), d_dest_[p], s_copy_);
    stage(flying, n * sizeof(int8_t), d_flying_[p], s_copy_);
    stage(weights, n * sizeof(double), d_weights_[p], s_copy_);
    if (groups && d_groups_[p])
      stage(groups, n * sizeof(uint16_t), d_groups_[p], s_copy_);
    if (responses) {
      if (!d_resp_[p]) d_resp_[p] = dmalloc<double>(n_ * nscores);
      stage(responses, n * nscores * sizeof(double), d_resp_[p], s_copy_);
    }
    PT_HIP_CHECK(hipEventRecord(copy_ev_[p], s_copy_));
    PT_HIP_CHECK(hipStreamWaitEvent(s_comp_, copy_ev_[p], 0));
    launch_move_chunks(origin ? d_origin_[p] : nullptr, d_dest_[p],
                       d_flying_[p], d_weights_[p],
                       groups ? d_groups_[p] : nullptr,
                       responses ? d_resp_[p] : nullptr, n, steps);
    PT_HIP_CHECK(hipEventRecord(kernels_done_[p], s_comp_));
    maybe_resort();
    // The caller may mutate or free its buffers as soon as move() returns
    // (the facade zeroes the flying array; transients die).  Block until
    // every H2D copy has consumed them; the walk keeps running async.
    PT_HIP_CHECK(hipStreamSynchronize(s_copy_));
    stats_.moves++;
  }

  void move_device(const double *d_origin, const double *d_dest,
                   const int8_t *d_flying, const double *d_weights, int64_t n,
                   const uint16_t *d_groups = nullptr,
                   const double *d_responses = nullptr) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    const int steps = max_steps > 0 ? max_steps : default_max_steps(mesh_);
    launch_move_chunks(d_origin, d_dest, d_flying, d_weights, d_groups,
                       d_responses, n, steps);
    maybe_resort();
    stats_.moves++;
  }

  // Persistent scratch for walk_raw.  The partitioned driver calls
  // walk_raw once per exchange round; per-call hipMalloc/hipFree pairs
  // would device-sync every round, so the buffers are grown once and
  // reused (freed in the dtor).
  struct WalkRawScratch {
    int64_t cap = 0;
    double *pos = nullptr, *dest = nullptr, *w = nullptr, *out_pos = nullptr,
           *resp = nullptr, *out_dest = nullptr;
    int32_t *elem = nullptr, *out_elem = nullptr;
    int8_t *status = nullptr;
    uint16_t *groups = nullptr;
    double *in_t = nullptr, *out_o = nullptr, *out_t = nullptr;
    int32_t *in_prev = nullptr, *out_prev = nullptr;
  };

  void ensure_wr_cap(int64_t n, bool need_groups, bool need_resp) {
    if (n > wr_.cap) {
      free_wr();
      wr_.cap = n + n / 4; // headroom against round-to-round growth
      wr_.pos = dmalloc<double>(wr_.cap * 3);
      wr_.dest = dmalloc<double>(wr_.cap * 3);
      wr_.w = dmalloc<double>(wr_.cap);
      wr_.out_pos = dmalloc<double>(wr_.cap * 3);
      wr_.elem = dmalloc<int32_t>(wr_.cap);
      wr_.out_elem = dmalloc<int32_t>(wr_.cap);
      wr_.status = dmalloc<int8_t>(wr_.cap);
    }
    if (need_groups && !wr_.groups) wr_.groups = dmalloc<uint16_t>(wr_.cap);
    if (need_resp && !wr_.resp) wr_.resp = dmalloc<double>(wr_.cap * nscores);
  }

  void free_wr() {
    for (void *q : {(void *)wr_.pos, (void *)wr_.dest, (void *)wr_.w,
                    (void *)wr_.out_pos, (void *)wr_.elem,
                    (void *)wr_.out_elem, (void *)wr_.status,
                    (void *)wr_.groups, (void *)wr_.resp,
                    (void *)wr_.out_dest, (void *)wr_.in_t,
                    (void *)wr_.out_o, (void *)wr_.out_t,
                    (void *)wr_.in_prev, (void *)wr_.out_prev})
      if (q) (void)hipFree(q);
    wr_ = WalkRawScratch{};
  }

  void walk_raw(int64_t n, const double *pos, const double *dest,
                const int32_t *elem, const double *weights, double *out_pos,
                int32_t *out_elem, int8_t *out_status,
                const uint16_t *groups = nullptr,
                const double *responses = nullptr,
                double *out_dest = nullptr, const double *in_t = nullptr,
                const int32_t *in_prev = nullptr, double *out_o = nullptr,
                double *out_t = nullptr,
                int32_t *out_prev = nullptr) override {
    if (n == 0) return;
    PT_HIP_CHECK(hipSetDevice(device_));
    const int steps = max_steps > 0 ? max_steps : default_max_steps(mesh_);
    ensure_wr_cap(n, groups != nullptr, responses != nullptr);
    const bool resume = in_t || in_prev || out_o || out_t || out_prev;
    if (resume && !wr_.in_t) {
      wr_.in_t = dmalloc<double>(wr_.cap);
      wr_.in_prev = dmalloc<int32_t>(wr_.cap);
      wr_.out_o = dmalloc<double>(wr_.cap * 3);
      wr_.out_t = dmalloc<double>(wr_.cap);
      wr_.out_prev = dmalloc<int32_t>(wr_.cap);
    }
    if (in_t)
      PT_HIP_CHECK(hipMemcpy(wr_.in_t, in_t, n * 8, hipMemcpyHostToDevice));
    if (in_prev)
      PT_HIP_CHECK(hipMemcpy(wr_.in_prev, in_prev, n * 4,
                             hipMemcpyHostToDevice));
    // Previous round's kernel has been synchronized below before this
    // call returns, so the scratch is free for reuse here.
    PT_HIP_CHECK(hipMemcpy(wr_.pos, pos, n * 3 * 8, hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(wr_.dest, dest, n * 3 * 8, hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(wr_.w, weights, n * 8, hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(wr_.elem, elem, n * 4, hipMemcpyHostToDevice));
    const uint16_t *dg = groups ? wr_.groups : nullptr;
    const double *dr = responses ? wr_.resp : nullptr;
    if (groups)
      PT_HIP_CHECK(hipMemcpy(wr_.groups, groups, n * 2, hipMemcpyHostToDevice));
    if (responses)
      PT_HIP_CHECK(hipMemcpy(wr_.resp, responses, n * nscores * 8,
                             hipMemcpyHostToDevice));
    if (out_dest && !wr_.out_dest) wr_.out_dest = dmalloc<double>(wr_.cap * 3);
    if (walk_fp32)
      k_walk_raw<true><<<grid_blocks(n), kBlock, 0, s_comp_>>>(
          d_planes_, d_planes32_, d_nbr_, wr_.pos, wr_.dest, wr_.elem, wr_.w,
          dg, dr, wr_.out_pos, wr_.out_elem, wr_.status,
          out_dest ? wr_.out_dest : nullptr, d_flux_, d_lost_,
          d_lostrec_, n,
          steps, reflective, d_face_bc_, ngroups, mesh_.nelems, nscores,
          d_pidx_, d_pelem_, d_pshift_, in_t ? wr_.in_t : nullptr,
          in_prev ? wr_.in_prev : nullptr, out_o ? wr_.out_o : nullptr,
          out_t ? wr_.out_t : nullptr, out_prev ? wr_.out_prev : nullptr);
    else
      k_walk_raw<false><<<grid_blocks(n), kBlock, 0, s_comp_>>>(
          d_planes_, d_planes32_, d_nbr_, wr_.pos, wr_.dest, wr_.elem, wr_.w,
          dg, dr, wr_.out_pos, wr_.out_elem, wr_.status,
          out_dest ? wr_.out_dest : nullptr, d_flux_, d_lost_,
          d_lostrec_, n,
          steps, reflective, d_face_bc_, ngroups, mesh_.nelems, nscores,
          d_pidx_, d_pelem_, d_pshift_, in_t ? wr_.in_t : nullptr,
          in_prev ? wr_.in_prev : nullptr, out_o ? wr_.out_o : nullptr,
          out_t ? wr_.out_t : nullptr, out_prev ? wr_.out_prev : nullptr);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
    PT_HIP_CHECK(hipMemcpy(out_pos, wr_.out_pos, n * 3 * 8,
                           hipMemcpyDeviceToHost));
    PT_HIP_CHECK(hipMemcpy(out_elem, wr_.out_elem, n * 4,
                           hipMemcpyDeviceToHost));
    PT_HIP_CHECK(hipMemcpy(out_status, wr_.status, n, hipMemcpyDeviceToHost));
    if (out_dest)
      PT_HIP_CHECK(hipMemcpy(out_dest, wr_.out_dest, n * 3 * 8,
                             hipMemcpyDeviceToHost));
    if (out_o)
      PT_HIP_CHECK(hipMemcpy(out_o, wr_.out_o, n * 3 * 8,
                             hipMemcpyDeviceToHost));
    if (out_t)
      PT_HIP_CHECK(hipMemcpy(out_t, wr_.out_t, n * 8,
                             hipMemcpyDeviceToHost));
    if (out_prev)
      PT_HIP_CHECK(hipMemcpy(out_prev, wr_.out_prev, n * 4,
                             hipMemcpyDeviceToHost));
  }

  void walk_raw_device(int64_t n, const double *d_pos, const double *d_dest,
                       const int32_t *d_elem, const double *d_weights,
                       double *d_out_pos, int32_t *d_out_elem,
                       int8_t *d_out_status,
                       const uint16_t *d_groups = nullptr,
                       const double *d_responses = nullptr,
                       double *d_out_dest = nullptr,
                       const double *d_in_t = nullptr,
                       const int32_t *d_in_prev = nullptr,
                       double *d_out_o = nullptr, double *d_out_t = nullptr,
                       int32_t *d_out_prev = nullptr) override {
    if (n == 0) return;
    PT_HIP_CHECK(hipSetDevice(device_));
    const int steps = max_steps > 0 ? max_steps : default_max_steps(mesh_);
    if (walk_fp32)
      k_walk_raw<true><<<grid_blocks(n), kBlock, 0, s_comp_>>>(
          d_planes_, d_planes32_, d_nbr_, d_pos, d_dest, d_elem, d_weights,
          d_groups, d_responses, d_out_pos, d_out_elem, d_out_status,
          d_out_dest, d_flux_, d_lost_, d_lostrec_, n, steps, reflective,
          d_face_bc_, ngroups, mesh_.nelems, nscores, d_pidx_, d_pelem_,
          d_pshift_, d_in_t, d_in_prev, d_out_o, d_out_t, d_out_prev);
    else
      k_walk_raw<false><<<grid_blocks(n), kBlock, 0, s_comp_>>>(
          d_planes_, d_planes32_, d_nbr_, d_pos, d_dest, d_elem, d_weights,
          d_groups, d_responses, d_out_pos, d_out_elem, d_out_status,
          d_out_dest, d_flux_, d_lost_, d_lostrec_, n, steps, reflective,
          d_face_bc_, ngroups, mesh_.nelems, nscores, d_pidx_, d_pelem_,
          d_pshift_, d_in_t, d_in_prev, d_out_o, d_out_t, d_out_prev);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
  }

  void end_batch() override {
    PT_HIP_CHECK(hipSetDevice(device_));
    sync();
    const int64_t fsz = fsz_;
    if (!d_bsum_) {
      d_bsum_ = dmalloc<double>(fsz);
      d_bsq_ = dmalloc<double>(fsz);
      PT_HIP_CHECK(hipMemset(d_bsum_, 0, fsz * sizeof(double)));
      PT_HIP_CHECK(hipMemset(d_bsq_, 0, fsz * sizeof(double)));
    }
    if (slices_ > 1) {
      k_reduce_slices<<<grid_blocks(fsz), kBlock, 0, s_comp_>>>(d_flux_, fsz,
                                                               slices_);
      PT_HIP_CHECK(hipGetLastError());
    }
    k_accumulate_batch<<<grid_blocks(fsz), kBlock, 0, s_comp_>>>(
        d_flux_, d_bsum_, d_bsq_, fsz);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
    nbatches_++;
  }
  std::vector<double> batch_sum() const override { return fetch_acc(d_bsum_); }
  std::vector<double> batch_sum_sq() const override { return fetch_acc(d_bsq_); }
  int64_t num_batches() const override { return nbatches_; }

  std::vector<double> fetch_acc(const double *p) const {
    const int64_t fsz = fsz_;
    std::vector<double> out(fsz, 0.0);
    if (p) {
      sync();
      PT_HIP_CHECK(hipMemcpy(out.data(), p, fsz * sizeof(double),
                             hipMemcpyDeviceToHost));
    }
    return out;
  }

  std::vector<double> flux() const override {
    sync();
    const int64_t fsz = fsz_;
    if (slices_ > 1) {
      k_reduce_slices<<<grid_blocks(fsz), kBlock, 0, s_comp_>>>(d_flux_, fsz,
                                                               slices_);
      PT_HIP_CHECK(hipGetLastError());
      PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
    }
    std::vector<double> out(fsz);
    PT_HIP_CHECK(hipMemcpy(out.data(), d_flux_, fsz * sizeof(double),
                           hipMemcpyDeviceToHost));
    return out;
  }

  std::vector<int32_t> elem_ids() const override {
    auto [s2c, elem] = fetch<int32_t>(d_elem_, 1);
    std::vector<int32_t> out(n_);
    for (int64_t i = 0; i < n_; ++i) out[s2c[i]] = elem[i];
    return out;
  }
  std::vector<double> positions() const override {
    auto [s2c, pos] = fetch<double>(d_pos_, 3);
    std::vector<double> out(n_ * 3);
    for (int64_t i = 0; i < n_; ++i)
      for (int k = 0; k < 3; ++k) out[(int64_t)s2c[i] * 3 + k] = pos[i * 3 + k];
    return out;
  }
  std::vector<uint8_t> escaped() const override {
    auto [s2c, esc] = fetch<uint8_t>(d_escaped_, 1);
    std::vector<uint8_t> out(n_);
    for (int64_t i = 0; i < n_; ++i) out[s2c[i]] = esc[i];
    return out;
  }

  const EngineStats &stats() const override {
    sync();
    unsigned long long lost = 0, loose = 0;
    PT_HIP_CHECK(hipMemcpy(&lost, d_lost_, sizeof lost, hipMemcpyDeviceToHost));
    PT_HIP_CHECK(
        hipMemcpy(&loose, d_loose_, sizeof loose, hipMemcpyDeviceToHost));
    stats_.lost_particles = (int64_t)lost;
    stats_.loose_localizations = (int64_t)loose;
    return stats_;
  }

  std::vector<double> lost_records() const override {
    sync();
    unsigned long long lost = 0;
    PT_HIP_CHECK(hipMemcpy(&lost, d_lost_, sizeof lost, hipMemcpyDeviceToHost));
    const int64_t k = std::min<int64_t>((int64_t)lost, kMaxLostRecords);
    std::vector<double> out((size_t)k * 4);
    if (k)
      PT_HIP_CHECK(hipMemcpy(out.data(), d_lostrec_, k * 4 * sizeof(double),
                             hipMemcpyDeviceToHost));
    return out;
  }

  void set_flux(const double *f, int64_t ne) override {
    if (ne != fsz_)
      throw std::runtime_error("set_flux size mismatch");
    sync();
    PT_HIP_CHECK(hipMemset(d_flux_, 0, ne * slices_ * sizeof(double)));
    PT_HIP_CHECK(hipMemcpy(d_flux_, f, ne * sizeof(double), hipMemcpyHostToDevice));
  }

  void set_particle_state(const double *pos, const int32_t *elem,
                          const uint8_t *escaped, int64_t n) override {
    check_n(n);
    PT_HIP_CHECK(hipSetDevice(device_));
    sync();
    // caller order -> identity map; next move's cadence re-sorts
    k_iota<<<grid_blocks(n_), kBlock, 0, s_comp_>>>(d_s2c_, n_);
    PT_HIP_CHECK(hipGetLastError());
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
    PT_HIP_CHECK(hipMemcpy(d_pos_, pos, n * 3 * sizeof(double), hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_elem_, elem, n * sizeof(int32_t), hipMemcpyHostToDevice));
    PT_HIP_CHECK(hipMemcpy(d_escaped_, escaped, n, hipMemcpyHostToDevice));
    moves_since_sort_ = sort_every_; // re-sort on the next move
  }

  void synchronize() override { sync(); }

  bool device_mesh(DeviceMeshView *out) const override {
    out->planes = d_planes_;
    out->nbr = d_nbr_;
    out->grid = grid_view_;
    out->stream = (void *)s_comp_;
    return true;
  }

private:
  void check_n(int64_t n) const {
    if (n != n_) throw std::runtime_error("particle count mismatch");
  }

  void sync() const {
    PT_HIP_CHECK(hipStreamSynchronize(s_copy_));
    PT_HIP_CHECK(hipStreamSynchronize(s_comp_));
  }

  // Async H2D from caller memory.  No implicit hipHostRegister: a
  // registration cache outliving freed caller buffers poisons the
  // runtime's pinning table for unrelated later copies.  Pinned sources
  // copy at full link rate; pageable sources take the runtime's internal
  // staging path -- always correct, just slower.
  void stage(const void *src, size_t bytes, void *dst, hipStream_t s) {
    PT_HIP_CHECK(hipMemcpyAsync(dst, src, bytes, hipMemcpyHostToDevice, s));
  }

  template <bool F32, bool Scored, bool Periodic, bool Agg = false>
  void launch_move_one(const double *origin, const double *dest,
                       const int8_t *flying, const double *weights,
                       const uint16_t *groups, const double *resp,
                       int64_t lo, int64_t hi, int steps) {
    k_move<F32, Scored, Periodic, Agg>
        <<<grid_blocks(hi - lo), kBlock, 0, s_comp_>>>(
        d_planes_, d_planes32_, d_nbr_, grid_view_, d_s2c_, origin, dest,
        flying, weights, groups, ngroups, d_pos_, d_elem_, d_escaped_,
        d_flux_, d_lost_, d_lostrec_, d_loose_, lo, hi, loc_tol_, steps,
        mesh_.nelems, slices_ - 1, reflective, d_face_bc_, resp, nscores,
        d_pidx_, d_pelem_, d_pshift_);
    PT_HIP_CHECK(hipGetLastError());
  }

  void launch_move_chunks(const double *origin, const double *dest,
                          const int8_t *flying, const double *weights,
                          const uint16_t *groups, const double *resp,
                          int64_t n, int steps) {
    // Chunked launches: a ~2.6M-slot launch keeps each XCD's Morton-
    // contiguous slot range's mesh working set inside its private L2.
    // Scored / periodic / wave-aggregated moves take dedicated k_move
    // instantiations; the plain (headline) instantiation compiles without
    // any of those features.
    const int64_t chunk = chunk_particles(n);
    const bool per = d_pidx_ != nullptr;
    // wave aggregation needs a single flat tally key per lane: plain
    // (unscored, ungrouped) moves only
    const bool agg = wave_agg() && !resp && !groups;
    for (int64_t lo = 0; lo < n; lo += chunk) {
      const int64_t hi = std::min(n, lo + chunk);
      auto go = [&](auto f32c, auto scoredc, auto perc, auto aggc) {
        launch_move_one<decltype(f32c)::value, decltype(scoredc)::value,
                        decltype(perc)::value, decltype(aggc)::value>(
            origin, dest, flying, weights, groups, resp, lo, hi, steps);
      };
      using T = std::true_type;
      using F = std::false_type;
      if (walk_fp32) {
        if (resp)     per ? go(T{}, T{}, T{}, F{}) : go(T{}, T{}, F{}, F{});
        else if (agg) per ? go(T{}, F{}, T{}, T{}) : go(T{}, F{}, F{}, T{});
        else          per ? go(T{}, F{}, T{}, F{}) : go(T{}, F{}, F{}, F{});
      } else {
        if (resp)     per ? go(F{}, T{}, T{}, F{}) : go(F{}, T{}, F{}, F{});
        else if (agg) per ? go(F{}, F{}, T{}, T{}) : go(F{}, F{}, F{}, T{});
        else          per ? go(F{}, F{}, T{}, F{}) : go(F{}, F{}, F{}, F{});
      }
    }
  }

  // Device Morton sort of particle slots by current position (hipCUB radix
  // sort); measured 2.2x on the walk for spatially random particle order.
  void spatial_sort() {
    k_morton_keys<<<grid_blocks(n_), kBlock, 0, s_comp_>>>(
        d_pos_, d_keys_, d_vals_, n_, mesh_.bbox_lo, sort_scale_);
    PT_HIP_CHECK(hipGetLastError());
    size_t bytes = sorttmp_bytes_;
    PT_HIP_CHECK(hipcub::DeviceRadixSort::SortPairs(
        d_sorttmp_, bytes, d_keys_, d_keys2_, d_vals_, d_order_, (int)n_, 0,
        64, s_comp_));
    k_permute_state<<<grid_blocks(n_), kBlock, 0, s_comp_>>>(
        d_order_, d_pos_, d_elem_, d_escaped_, d_s2c_, d_pos2_, d_elem2_,
        d_esc2_, d_s2c2_, n_);
    PT_HIP_CHECK(hipGetLastError());
    std::swap(d_pos_, d_pos2_);
    std::swap(d_elem_, d_elem2_);
    std::swap(d_escaped_, d_esc2_);
    std::swap(d_s2c_, d_s2c2_);
  }

  void maybe_resort() {
    if (sort_every_ > 0 && ++moves_since_sort_ >= sort_every_) {
      spatial_sort();
      moves_since_sort_ = 0;
    }
  }

  // D2H of (s2c, state-array) for caller-order readback.
  template <class T>
  std::pair<std::vector<int32_t>, std::vector<T>> fetch(const T *dev,
                                                        int comps) const {
    sync();
    std::vector<int32_t> s2c(n_);
    std::vector<T> v(n_ * comps);
    PT_HIP_CHECK(hipMemcpy(s2c.data(), d_s2c_, n_ * sizeof(int32_t),
                           hipMemcpyDeviceToHost));
    PT_HIP_CHECK(hipMemcpy(v.data(), dev, n_ * comps * sizeof(T),
                           hipMemcpyDeviceToHost));
    return {std::move(s2c), std::move(v)};
  }

  Mesh mesh_;
  int64_t n_;
  int64_t fsz_ = 0; // nelems * ngroups * nscores
  int device_ = 0;
  double loc_tol_ = 1e-12;
  int slices_ = 1;
  int sort_every_ = 0;
  int64_t moves_since_sort_ = 0;
  uint64_t parity_ = 0;
  Vec3 sort_scale_{0, 0, 0};

  hipStream_t s_copy_{}, s_comp_{};
  std::array<hipEvent_t, 2> copy_ev_{};
  std::array<hipEvent_t, 2> kernels_done_{};

  Plane *d_planes_ = nullptr;
  Plane32 *d_planes32_ = nullptr;
  int32_t *d_nbr_ = nullptr;
  int32_t *d_cell_start_ = nullptr;
  int32_t *d_cell_tets_ = nullptr;
  GridView grid_view_{};

  double *d_pos_ = nullptr;
  int32_t *d_elem_ = nullptr;
  uint8_t *d_escaped_ = nullptr;
  int32_t *d_s2c_ = nullptr;
  double *d_flux_ = nullptr;
  unsigned long long *d_lost_ = nullptr;
  unsigned long long *d_loose_ = nullptr;
  double *d_lostrec_ = nullptr;
  double *d_origin_[2] = {nullptr, nullptr};
  uint16_t *d_groups_[2] = {nullptr, nullptr};
  double *d_resp_[2] = {nullptr, nullptr};
  double *d_bsum_ = nullptr, *d_bsq_ = nullptr;
  uint32_t *d_face_bc_ = nullptr;
  int32_t *d_pidx_ = nullptr, *d_pelem_ = nullptr;
  double *d_pshift_ = nullptr;
  WalkRawScratch wr_;
  int64_t nbatches_ = 0;
  double *d_dest_[2] = {nullptr, nullptr};
  int8_t *d_flying_[2] = {nullptr, nullptr};
  double *d_weights_[2] = {nullptr, nullptr};

  uint64_t *d_keys_ = nullptr, *d_keys2_ = nullptr;
  int32_t *d_vals_ = nullptr, *d_order_ = nullptr;
  double *d_pos2_ = nullptr;
  int32_t *d_elem2_ = nullptr;
  uint8_t *d_esc2_ = nullptr;
  int32_t *d_s2c2_ = nullptr;
  void *d_sorttmp_ = nullptr;
  size_t sorttmp_bytes_ = 0;

  mutable EngineStats stats_;
};

} // namespace

std::unique_ptr<Engine> make_gpu_engine(Mesh mesh, int64_t num_particles,
                                        int device, int ngroups, int nscores) {
  int count = 0;
  if (hipGetDeviceCount(&count) != hipSuccess || count <= device) {
    (void)hipGetLastError();
    return nullptr;
  }
  return std::make_unique<GpuEngine>(std::move(mesh), num_particles, device,
                                     ngroups, nscores);
}

} // namespace pumitally
