// The adjacency walk: the single hot algorithm of the framework.
//
// Re-implements from scratch the behavior of the external
// pumipic::ParticleTracer element walk that the reference delegates to
// (/root/reference/src/pumitally/PumiTallyImpl.cpp:454 and the handler
// sequencing at :297-316): walk a straight segment origin->dest through the
// tet mesh element by element, accumulate track_length*weight into the flux
// tally of every element crossed (K9, :352-380), clip the destination at a
// vacuum boundary (K6, :256-286), and advance the current element (K5,
// :243-254) -- all fused into one loop per particle instead of one kernel
// per walk iteration over all particles.
//
// The exit-face test is a signed-distance parametric test against the 4
// precomputed inward-positive face planes: V(t) = plane_eval(p(t)) is linear
// in t, the particle exits through the face with the smallest crossing
// t = V_o / (V_o - V_d) among faces with V_d < 0.  Because shared-face
// planes are bitwise-consistent across neighbor tets (mesh.cpp finalize()
// step 3), t is monotone and the per-element intervals tile the segment
// exactly.
//
// Header-only, host+device: the CPU engine is the oracle for the HIP kernel.
#pragma once

#include "geom.h"

namespace pumitally {

constexpr double kWalkTEps = 1e-12;   // tolerance on the segment parameter t
// out_elem value when max_steps hit.  Must not collide with the encoded
// foreign-element refs -(2+k) used by partitioned submeshes.
constexpr int32_t kWalkLost = INT32_MIN;

// Walk state for the incremental (one-crossing-at-a-time) API.  The GPU
// move kernel uses this so a lane whose walk finished can immediately
// acquire its next particle instead of idling behind the wave's slowest
// lane (variable walk lengths gave 46% VALUUtilization with the
// whole-segment-per-lane formulation).
struct WalkState {
  Vec3 o, d;
  Vec3f of, df; // fp32 copies for the fp32-traversal fast path
  double seg_len, t_cur, weight;
  int32_t elem, prev_elem;
  int step;
  int wraps; // periodic-translation restarts taken (see kMaxWraps)
  bool tally;
};

// A periodic restart resets the per-wrap step budget (a long segment
// wrapping a small periodic box many times is geometrically valid and must
// not be dropped as lost), but the number of wraps itself is capped so a
// numerically stuck zero-progress bounce between periodic faces still
// terminates.  4096 wraps is far beyond any physical chord; combined with
// max_steps per wrap the worst-case per-particle iteration count stays
// bounded.
constexpr int kMaxWraps = 4096;

PT_HD void walk_init(WalkState &s, int32_t elem, Vec3 o, Vec3 d, double w) {
  s.o = o;
  s.d = d;
  s.of = Vec3f{(float)o.x, (float)o.y, (float)o.z};
  s.df = Vec3f{(float)d.x, (float)d.y, (float)d.z};
  s.seg_len = norm(d - o);
  s.t_cur = 0.0;
  s.weight = w;
  s.elem = elem;
  s.prev_elem = -1;
  s.step = 0;
  s.wraps = 0;
  s.tally = (w != 0.0) && (s.seg_len > 0.0);
}

// Advance one element crossing.  Returns true when the walk finished and
// the out_* values are valid.  FluxAdd: void(int32_t elem, double v).
// Specular reflection of point p across plane pl (unit normal).
PT_HD Vec3 reflect_point(const Plane &pl, Vec3 p) {
  const double v = plane_eval(pl, p);
  return {p.x - 2.0 * v * pl.nx, p.y - 2.0 * v * pl.ny, p.z - 2.0 * v * pl.nz};
}

// Periodic-translation helper shared by both advance variants: teleport the
// remaining segment by the pair's translation vector and resume in the
// paired element (a translation is an isometry, so total tallied length is
// conserved, exactly like the reflective restart).
PT_HD void periodic_restart(WalkState &s, double t_clamped,
                                   int32_t pair_elem, const double *shift) {
  const Vec3 hit = s.o + t_clamped * (s.d - s.o);
  const Vec3 T{shift[0], shift[1], shift[2]};
  const double remaining = (1.0 - t_clamped) * s.seg_len;
  s.o = hit + T;
  s.d = s.d + T;
  s.of = Vec3f{(float)s.o.x, (float)s.o.y, (float)s.o.z};
  s.df = Vec3f{(float)s.d.x, (float)s.d.y, (float)s.d.z};
  s.seg_len = remaining > 0.0 ? remaining : 0.0;
  s.t_cur = 0.0;
  s.prev_elem = -1;
  s.elem = pair_elem;
  s.step = 0; // fresh per-wrap budget; total wraps capped by kMaxWraps
  s.wraps++;
}

// Periodic=false (the default, and the headline GPU instantiation) compiles
// the pairing arguments away entirely -- codegen identical to the
// pre-periodic kernel.  Periodic=true callers may still pass pidx=nullptr
// (no pairs on this mesh); the check is null-safe.
template <bool Periodic = false, class FluxAdd>
PT_HD bool walk_advance(const Plane *__restrict__ planes,
                        const int32_t *__restrict__ nbr, WalkState &s,
                        int max_steps, FluxAdd &&add, int32_t *out_elem,
                        Vec3 *out_pos, bool *out_escaped,
                        bool reflective = false,
                        const uint32_t *__restrict__ face_bc = nullptr,
                        const int32_t *__restrict__ pidx = nullptr,
                        const int32_t *__restrict__ pelem = nullptr,
                        const double *__restrict__ pshift = nullptr) {
  bool budget_exhausted = s.step++ >= max_steps;
  if constexpr (Periodic) budget_exhausted |= s.wraps >= kMaxWraps;
  if (budget_exhausted) {
    // Did not converge (numerically stuck / absurd chord): drop here and
    // flag as lost (reference prints "Not all particles are found",
    // PumiTallyImpl.cpp:455-458).
    *out_elem = kWalkLost;
    *out_pos = s.o + s.t_cur * (s.d - s.o);
    *out_escaped = false;
    return true;
  }
  // Evaluate the 4 face planes at both segment endpoints.  Candidate
  // comparisons use cross-multiplied forms (all denominators positive):
  //   t_f >= t_cur - eps  <=>  vo >= (t_cur - eps) * den
  //   t_f <  t_best       <=>  vo * den_best < num_best * den
  // so the fp64 division (40-60 cycles) happens ONCE for the chosen face
  // instead of once per candidate.
  const Plane *pl = planes + (int64_t)s.elem * 4;
  double num_best = 2.0, den_best = 1.0; // t_best = 2 (no candidate yet)
  int exit_face = -1;
  const double t_floor = s.t_cur - kWalkTEps;
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int f = 0; f < 4; ++f) {
    const double vd = plane_eval(pl[f], s.d);
    if (vd < 0.0) {
      const double vo = plane_eval(pl[f], s.o);
      const double denom = vo - vd; // > 0 since vd < 0 <= ~vo
      if (denom > 0.0) {
        // Never step backwards (grazing entry can give t slightly below
        // t_cur); never consider the face we just came through.
        if (nbr[(int64_t)s.elem * 4 + f] != s.prev_elem || s.prev_elem == -1) {
          if (vo >= t_floor * denom && vo * den_best < num_best * denom) {
            num_best = vo;
            den_best = denom;
            exit_face = f;
          }
        }
      }
    }
  }
  const double t_exit = exit_face >= 0 ? num_best / den_best : 2.0;

  if (exit_face < 0 || t_exit >= 1.0) {
    // Destination lies inside this element: tally the final partial
    // segment and stop (reference: reached_destination, last_exit==-1).
    if (s.tally) add(s.elem, (1.0 - s.t_cur) * s.seg_len * s.weight);
    *out_elem = s.elem;
    *out_pos = s.d;
    *out_escaped = false;
    return true;
  }

  const double t_clamped = t_exit > s.t_cur ? t_exit : s.t_cur;
  if (s.tally) add(s.elem, (t_clamped - s.t_cur) * s.seg_len * s.weight);

  const int32_t next = nbr[(int64_t)s.elem * 4 + exit_face];
  if (next == -1) {
    const int64_t fidx = (int64_t)s.elem * 4 + exit_face;
    if constexpr (Periodic) {
      if (pidx && pidx[fidx] >= 0) {
        const int32_t pk = pidx[fidx];
        periodic_restart(s, t_clamped, pelem[pk], pshift + (int64_t)pk * 3);
        return false;
      }
    }
    const bool refl_here =
        reflective ||
        (face_bc && ((face_bc[fidx >> 5] >> (fidx & 31)) & 1u));
    if (refl_here) {
      // Specular reflection: restart the segment at the crossing with the
      // remaining part mirrored across the boundary plane (an isometry, so
      // total tallied length is conserved).  The incoming face plane sees
      // the mirrored destination on its positive side, so it cannot be
      // re-selected.
      const Vec3 hit = s.o + t_clamped * (s.d - s.o);
      const Vec3 d2 = reflect_point(pl[exit_face], s.d);
      const double remaining = (1.0 - t_clamped) * s.seg_len;
      s.o = hit;
      s.d = d2;
      s.of = Vec3f{(float)hit.x, (float)hit.y, (float)hit.z};
      s.df = Vec3f{(float)d2.x, (float)d2.y, (float)d2.z};
      s.seg_len = remaining > 0.0 ? remaining : 0.0;
      s.t_cur = 0.0;
      s.prev_elem = -1;
      return false;
    }
    // Vacuum boundary: clip the destination to the exit point; the
    // particle keeps its last element id (reference K6 semantics,
    // PumiTallyImpl.cpp:275-281 and the 1.0-not-1.1 test expectation).
    *out_elem = s.elem;
    *out_pos = s.o + t_clamped * (s.d - s.o);
    *out_escaped = true;
    return true;
  }
  if (next < -1) {
    // Partitioned submesh: the face crosses into an element owned by
    // another rank.  Stop at the crossing; the caller decodes the
    // foreign reference (k = -(next+2)) and ships the particle.  The
    // state keeps (s.o, s.t_cur, s.elem) so the handoff record can carry
    // the walk's t-parametrization and the receiver replays the
    // remaining crossings with the SAME fp decisions as an uncut walk
    // (bitwise-identical element attribution; see walk_segment's
    // resume_t/resume_prev).
    s.t_cur = t_clamped;
    *out_elem = next;
    *out_pos = s.o + t_clamped * (s.d - s.o);
    *out_escaped = false;
    return true;
  }
  s.prev_elem = s.elem;
  s.elem = next;
  s.t_cur = t_clamped;
  return false;
}

// fp32-traversal variant: exit-face CANDIDATE selection runs on the fp32
// planes (half the footprint, 2x the VALU rate -- CDNA4 fp64 vector rate is
// half of fp32); the CHOSEN face's crossing parameter t is then recomputed
// from the fp64 plane with the fp64 endpoints, so every tallied interval
// and every committed position is fp64-exact and the intervals still
// telescope to the full segment length (conservation holds to 1e-15).
// A near-tie mis-ordering of candidate faces only re-routes the walk
// through the neighbor for an O(1e-7)-long sliver, self-healed by the
// monotone-t clamp.
constexpr float kWalkTEps32 = 1e-6f;

template <bool Periodic = false, class FluxAdd>
PT_HD bool walk_advance32(const Plane *__restrict__ planes,
                          const Plane32 *__restrict__ planes32,
                          const int32_t *__restrict__ nbr, WalkState &s,
                          int max_steps, FluxAdd &&add, int32_t *out_elem,
                          Vec3 *out_pos, bool *out_escaped,
                          bool reflective = false,
                          const uint32_t *__restrict__ face_bc = nullptr,
                          const int32_t *__restrict__ pidx = nullptr,
                          const int32_t *__restrict__ pelem = nullptr,
                          const double *__restrict__ pshift = nullptr) {
  bool budget_exhausted = s.step++ >= max_steps;
  if constexpr (Periodic) budget_exhausted |= s.wraps >= kMaxWraps;
  if (budget_exhausted) {
    *out_elem = kWalkLost;
    *out_pos = s.o + s.t_cur * (s.d - s.o);
    *out_escaped = false;
    return true;
  }
  const Plane32 *pl = planes32 + (int64_t)s.elem * 4;
  const float tcur_f = (float)s.t_cur;
  float t_exit = 2.0f;
  int exit_face = -1;
#if defined(__HIP_DEVICE_COMPILE__)
#pragma unroll
#endif
  for (int f = 0; f < 4; ++f) {
    const float vd = plane_eval32(pl[f], s.df);
    if (vd < 0.0f) {
      const float vo = plane_eval32(pl[f], s.of);
      const float denom = vo - vd;
      if (denom > 0.0f) {
        const float tf = vo / denom;
        if (nbr[(int64_t)s.elem * 4 + f] != s.prev_elem || s.prev_elem == -1) {
          if (tf >= tcur_f - kWalkTEps32 && tf < t_exit) {
            t_exit = tf;
            exit_face = f;
          }
        }
      }
    }
  }

  if (exit_face < 0 || t_exit >= 1.0f) {
    if (s.tally) add(s.elem, (1.0 - s.t_cur) * s.seg_len * s.weight);
    *out_elem = s.elem;
    *out_pos = s.d;
    *out_escaped = false;
    return true;
  }

  // Exact fp64 crossing of the chosen face.
  const Plane &pe = planes[(int64_t)s.elem * 4 + exit_face];
  const double vo64 = plane_eval(pe, s.o);
  const double vd64 = plane_eval(pe, s.d);
  double t64 = (vo64 - vd64) > 0.0 ? vo64 / (vo64 - vd64) : s.t_cur;
  if (t64 > 1.0) t64 = 1.0;
  const double t_clamped = t64 > s.t_cur ? t64 : s.t_cur;
  if (s.tally) add(s.elem, (t_clamped - s.t_cur) * s.seg_len * s.weight);

  const int32_t next = nbr[(int64_t)s.elem * 4 + exit_face];
  if (next == -1) {
    const int64_t fidx = (int64_t)s.elem * 4 + exit_face;
    if constexpr (Periodic) {
      if (pidx && pidx[fidx] >= 0) {
        const int32_t pk = pidx[fidx];
        periodic_restart(s, t_clamped, pelem[pk], pshift + (int64_t)pk * 3);
        return false;
      }
    }
    const bool refl_here =
        reflective ||
        (face_bc && ((face_bc[fidx >> 5] >> (fidx & 31)) & 1u));
    if (refl_here) {
      const Plane &pe64 = planes[(int64_t)s.elem * 4 + exit_face];
      const Vec3 hit = s.o + t_clamped * (s.d - s.o);
      const Vec3 d2 = reflect_point(pe64, s.d);
      const double remaining = (1.0 - t_clamped) * s.seg_len;
      s.o = hit;
      s.d = d2;
      s.of = Vec3f{(float)hit.x, (float)hit.y, (float)hit.z};
      s.df = Vec3f{(float)d2.x, (float)d2.y, (float)d2.z};
      s.seg_len = remaining > 0.0 ? remaining : 0.0;
      s.t_cur = 0.0;
      s.prev_elem = -1;
      return false;
    }
    *out_elem = s.elem;
    *out_pos = s.o + t_clamped * (s.d - s.o);
    *out_escaped = true;
    return true;
  }
  if (next < -1) {
    s.t_cur = t_clamped; // export resume state (see fp64 variant)
    *out_elem = next;
    *out_pos = s.o + t_clamped * (s.d - s.o);
    *out_escaped = false;
    return true;
  }
  s.prev_elem = s.elem;
  s.elem = next;
  s.t_cur = t_clamped;
  return false;
}

// FluxAdd: functor void(int32_t elem, double contribution).  On the GPU this
// performs atomicAdd into the flux array; on the serial CPU path a plain +=.
// out_dest (optional): the walk's CURRENT destination at termination.
// Reflective restarts mirror the destination and periodic restarts
// translate it, so after either a handoff record must ship s.d -- not
// the caller's original d -- or the receiving rank walks to a stale
// point (the round-2 periodic-partitioned ping-pong bug).
//
// Bitwise-exact handoff resume (resume_t / resume_prev in, out_o / out_t
// / out_prev out): the walk is t-parametrized over (o, d), so a
// cut-crossing record that carries the CURRENT wrap-segment origin s.o
// (as its position), the progress t at the crossing, and the element
// exited from lets the receiving rank continue with s.t_cur = t and
// s.prev_elem = that element -- every subsequent plane evaluation,
// monotone-t clamp and exit-face choice then reproduces the uncut
// walk's fp decisions bit for bit, and partitioned flux attribution is
// ELEMENTWISE identical to the replicated engine's (not merely
// conservative).  Without this, a resumed walk re-based at the crossing
// point computes crossings from perturbed endpoints, and a track
// passing within fp noise of a face-edge junction can attribute its
// final sliver to the neighboring tet (found by tools/part_world2_soak
// at ~1 per 10^5 handoffs; conservation still held).
template <bool Periodic = false, class FluxAdd>
PT_HD void walk_segment(const Plane *__restrict__ planes,
                        const int32_t *__restrict__ nbr, int32_t elem, Vec3 o,
                        Vec3 d, double weight, int max_steps, FluxAdd &&add,
                        int32_t *out_elem, Vec3 *out_pos, bool *out_escaped,
                        bool reflective = false,
                        const uint32_t *__restrict__ face_bc = nullptr,
                        const int32_t *__restrict__ pidx = nullptr,
                        const int32_t *__restrict__ pelem = nullptr,
                        const double *__restrict__ pshift = nullptr,
                        Vec3 *out_dest = nullptr, double resume_t = 0.0,
                        int32_t resume_prev = -1, Vec3 *out_o = nullptr,
                        double *out_t = nullptr,
                        int32_t *out_prev = nullptr) {
  WalkState s;
  walk_init(s, elem, o, d, weight);
  s.t_cur = resume_t;
  s.prev_elem = resume_prev;
  while (!walk_advance<Periodic>(planes, nbr, s, max_steps, add, out_elem,
                                 out_pos, out_escaped, reflective, face_bc,
                                 pidx, pelem, pshift)) {
  }
  if (out_dest) *out_dest = s.d;
  if (out_o) *out_o = s.o;
  if (out_t) *out_t = s.t_cur;
  if (out_prev) *out_prev = s.elem; // element exited from, on a handoff
}

template <bool Periodic = false, class FluxAdd>
PT_HD void walk_segment32(const Plane *__restrict__ planes,
                          const Plane32 *__restrict__ planes32,
                          const int32_t *__restrict__ nbr, int32_t elem,
                          Vec3 o, Vec3 d, double weight, int max_steps,
                          FluxAdd &&add, int32_t *out_elem, Vec3 *out_pos,
                          bool *out_escaped, bool reflective = false,
                          const uint32_t *__restrict__ face_bc = nullptr,
                          const int32_t *__restrict__ pidx = nullptr,
                          const int32_t *__restrict__ pelem = nullptr,
                          const double *__restrict__ pshift = nullptr,
                          Vec3 *out_dest = nullptr, double resume_t = 0.0,
                          int32_t resume_prev = -1, Vec3 *out_o = nullptr,
                          double *out_t = nullptr,
                          int32_t *out_prev = nullptr) {
  WalkState s;
  walk_init(s, elem, o, d, weight);
  s.t_cur = resume_t;
  s.prev_elem = resume_prev;
  while (!walk_advance32<Periodic>(planes, planes32, nbr, s, max_steps, add,
                                   out_elem, out_pos, out_escaped, reflective,
                                   face_bc, pidx, pelem, pshift)) {
  }
  if (out_dest) *out_dest = s.d;
  if (out_o) *out_o = s.o;
  if (out_t) *out_t = s.t_cur;
  if (out_prev) *out_prev = s.elem;
}

// Point-in-tet test against the 4 planes (signed distances, unit normals).
PT_HD bool tet_contains(const Plane *__restrict__ planes, int32_t t, Vec3 p,
                        double tol) {
  const Plane *pl = planes + (int64_t)t * 4;
  double mn = plane_eval(pl[0], p);
  mn = mn < plane_eval(pl[1], p) ? mn : plane_eval(pl[1], p);
  mn = mn < plane_eval(pl[2], p) ? mn : plane_eval(pl[2], p);
  mn = mn < plane_eval(pl[3], p) ? mn : plane_eval(pl[3], p);
  return mn >= -tol;
}

// Grid-based localization shared by CPU and GPU.  Mirrors Mesh::locate but
// takes flat arrays so the HIP kernel can use it directly.
struct GridView {
  int nx, ny, nz;
  Vec3 lo, inv_h;
  const int32_t *cell_start;
  const int32_t *cell_tets;
};

// used_loose (optional): set to true when only the tol*1e4 retry pass
// succeeded (the strict pass can fail by a few ulps for points exactly on
// faces); callers count these into EngineStats::loose_localizations.
PT_HD int32_t grid_locate(const GridView &g, const Plane *__restrict__ planes,
                          Vec3 p, double tol, bool *used_loose = nullptr) {
  int cx = (int)((p.x - g.lo.x) * g.inv_h.x);
  int cy = (int)((p.y - g.lo.y) * g.inv_h.y);
  int cz = (int)((p.z - g.lo.z) * g.inv_h.z);
  cx = cx < 0 ? 0 : (cx >= g.nx ? g.nx - 1 : cx);
  cy = cy < 0 ? 0 : (cy >= g.ny ? g.ny - 1 : cy);
  cz = cz < 0 ? 0 : (cz >= g.nz ? g.nz - 1 : cz);
  const int64_t c = ((int64_t)cz * g.ny + cy) * g.nx + cx;
  const int32_t b = g.cell_start[c], e = g.cell_start[c + 1];
  for (int32_t i = b; i < e; ++i)
    if (tet_contains(planes, g.cell_tets[i], p, tol)) return g.cell_tets[i];
  for (int32_t i = b; i < e; ++i)
    if (tet_contains(planes, g.cell_tets[i], p, tol * 1e4)) {
      if (used_loose) *used_loose = true;
      return g.cell_tets[i];
    }
  return -1;
}

} // namespace pumitally
