// Tally engine interface: the device-agnostic contract behind the public
// PumiTally API.  Two implementations:
//   * CpuEngine  (engine_cpu.cpp)  - serial oracle, runs everywhere
//   * GpuEngine  (../hip/engine_gpu.hip) - MI355X HIP engine
//
// Semantics mirror the reference's 4-call flow
// (/root/reference/src/pumitally/PumiTally.h:50-103):
//   ctor                 -> all particles at the centroid of element 0
//                           (PumiTallyImpl.cpp:492-528)
//   copy_initial_position-> localize each particle at its given position;
//                           no tallying (PumiTallyImpl.cpp:54-64,195-221)
//   move                 -> phase A: relocate flying particles to the given
//                           origin WITHOUT tallying; phase B: walk them to
//                           the destination, tallying track_length*weight
//                           per element crossed (PumiTallyImpl.cpp:66-149)
//   write_tally_results  -> normalize flux by element volume, write VTK
//                           (PumiTallyImpl.cpp:151-157,382-416)
//
// Behavioral pins taken from the reference tests
// (test/test_pumi_tally_impl_methods.cpp):
//   * a particle that exited through the vacuum boundary keeps its clipped
//     position and element id; phase A does NOT relocate it (the 2nd-move
//     flux expectations at :361-389 are only satisfiable this way)
//   * non-flying particles do not move and do not tally
//   * flux accumulates across move calls until write
//
// Design deviation (MI355X-first): phase A is a direct grid localization of
// the particles whose origin actually changed, not a zero-weight walk
// through the mesh as in the reference -- same observable state, a fraction
// of the work.
#pragma once

#include "mesh.h"
#include "walk.h"

#include <cstdint>
#include <cstdlib>
#include <stdexcept>
#include <memory>
#include <string>
#include <vector>

namespace pumitally {

struct EngineStats {
  int64_t lost_particles = 0;   // walks that hit max_steps
  int64_t moves = 0;            // move() calls
  int64_t relocated = 0;        // phase-A relocations performed
  // Localizations that only succeeded at the loosened tolerance (tol*1e4
  // retry in grid_locate / Mesh::locate).  A nonzero count near partition
  // cuts can mean mis-assigned particles; surfaced so it is never silent.
  int64_t loose_localizations = 0;
};

// First-K lost-particle capture (walks that hit max_steps): enough to find
// and reproduce the offending histories the day a real mesh produces a
// nonzero lost count, without any steady-state cost.
constexpr int kMaxLostRecords = 16;

class Engine {
public:
  virtual ~Engine() = default;

  virtual int64_t num_particles() const = 0;
  virtual const Mesh &mesh() const = 0;

  // positions: n*3 doubles (x,y,z interleaved).
  virtual void copy_initial_position(const double *positions, int64_t n) = 0;

  // origin/dest: n*3 doubles; flying: n int8; weights: n doubles.
  // groups (optional, nullable): per-particle energy-group index in
  // [0, ngroups); contributions land in flux[group*nelems + elem].  The
  // reference has a single scalar tally; ngroups=1 (the default) matches
  // it exactly.
  // responses (optional, nullable): n*nscores per-particle response
  // multipliers for simultaneous tally SCORES (e.g. flux + heating +
  // fission from one walk).  With responses given, score k of a crossing
  // tallies seg * weight * responses[i*nscores+k] into
  // flux[(k*ngroups + group)*nelems + elem].  With responses == nullptr
  // only score 0 is tallied (plain seg * weight); scores 1..nscores-1
  // stay zero (see test_scored_null_responses_is_plain_flux).  nscores=1
  // with null responses (the default) is exactly the reference's single
  // tally.
  virtual void move(const double *origin, const double *dest,
                    const int8_t *flying, const double *weights, int64_t n,
                    const uint16_t *groups = nullptr,
                    const double *responses = nullptr) = 0;

  // Fast path for callers that know no particle was resampled this step
  // (origin == committed position for every particle): skips the origin
  // upload and phase A entirely.  origin=nullptr in move() semantics.
  virtual void move_continue(const double *dest, const int8_t *flying,
                             const double *weights, int64_t n,
                             const uint16_t *groups = nullptr,
                             const double *responses = nullptr) {
    move(nullptr, dest, flying, weights, n, groups, responses);
  }

  // Device-resident move: all arrays already live in this engine's device
  // memory (for GPU-side transport codes; no host staging at all).
  // origin may be nullptr (continue semantics).  Throws on the CPU engine.
  virtual void move_device(const double *d_origin, const double *d_dest,
                           const int8_t *d_flying, const double *d_weights,
                           int64_t n, const uint16_t *d_groups = nullptr,
                           const double *d_responses = nullptr) {
    (void)d_origin; (void)d_dest; (void)d_flying; (void)d_weights; (void)n;
    (void)d_groups; (void)d_responses;
    throw std::runtime_error("move_device requires the GPU engine");
  }

  // Raw batched walk for the partitioned driver: walks n independent
  // segments (pos->dest) starting in the given elements, tallying into
  // this engine's flux.  status: 0=reached dest, 1=escaped (vacuum),
  // 2=handoff (out_elem = encoded foreign ref -(2+k)), 3=lost.
  // groups (nullable): per-segment energy-group index, same semantics as
  // move(); a handed-off particle keeps its group (the partitioned driver
  // carries it in the exchange record).  responses (nullable): n*nscores
  // per-segment score multipliers, same semantics as move().
  // Synchronous; host memory.
  // out_dest (optional, n*3): the walk's CURRENT destination at
  // termination -- reflective restarts mirror it, periodic restarts
  // translate it, so a handoff (status 2) must resume toward out_dest,
  // not the original dest.
  // Bitwise-exact handoff resume (walk.h walk_segment doc): in_t/in_prev
  // (optional, n) seed the walk's segment progress and entry-exclusion
  // element for resumed particles; out_o (n*3) / out_t (n) / out_prev
  // (n) export the wrap-segment origin, the progress t at termination,
  // and the element a handoff exited from.  A handoff record carrying
  // (out_o, out_dest, out_t, out_prev) lets the receiving rank replay
  // the remaining crossings with fp decisions identical to an uncut
  // walk, so partitioned flux attribution matches the replicated
  // engine elementwise.
  virtual void walk_raw(int64_t n, const double *pos, const double *dest,
                        const int32_t *elem, const double *weights,
                        double *out_pos, int32_t *out_elem,
                        int8_t *out_status,
                        const uint16_t *groups = nullptr,
                        const double *responses = nullptr,
                        double *out_dest = nullptr,
                        const double *in_t = nullptr,
                        const int32_t *in_prev = nullptr,
                        double *out_o = nullptr, double *out_t = nullptr,
                        int32_t *out_prev = nullptr) = 0;

  // Device-resident walk_raw: every array already lives in this engine's
  // device memory (no staging at all -- the partitioned driver keeps its
  // round loop on device and exchanges records over RCCL directly).
  // Synchronous like walk_raw.  Throws on the CPU engine.
  virtual void walk_raw_device(int64_t n, const double *d_pos,
                               const double *d_dest, const int32_t *d_elem,
                               const double *d_weights, double *d_out_pos,
                               int32_t *d_out_elem, int8_t *d_out_status,
                               const uint16_t *d_groups = nullptr,
                               const double *d_responses = nullptr,
                               double *d_out_dest = nullptr,
                               const double *d_in_t = nullptr,
                               const int32_t *d_in_prev = nullptr,
                               double *d_out_o = nullptr,
                               double *d_out_t = nullptr,
                               int32_t *d_out_prev = nullptr) {
    (void)n; (void)d_pos; (void)d_dest; (void)d_elem; (void)d_weights;
    (void)d_out_pos; (void)d_out_elem; (void)d_out_status; (void)d_groups;
    (void)d_responses; (void)d_out_dest; (void)d_in_t; (void)d_in_prev;
    (void)d_out_o; (void)d_out_t; (void)d_out_prev;
    throw std::runtime_error("walk_raw_device requires the GPU engine");
  }

  // Read back state (host copies).
  virtual std::vector<double> flux() const = 0;           // nelems*ngroups*nscores, raw tally
  virtual std::vector<int32_t> elem_ids() const = 0;      // n
  virtual std::vector<double> positions() const = 0;      // n*3
  virtual std::vector<uint8_t> escaped() const = 0;       // n

  virtual const EngineStats &stats() const = 0;

  // First min(lost_particles, kMaxLostRecords) lost-walk records, 4 doubles
  // each: (caller particle index -- or segment index for walk_raw --,
  // drop x, drop y, drop z).  Empty when nothing was lost.
  virtual std::vector<double> lost_records() const { return {}; }

  // Batch statistics (standard MC uncertainty accounting; the reference
  // has a single accumulating tally with no variance).  end_batch()
  // accumulates the current flux into running sum / sum-of-squares and
  // zeroes the per-batch tally; batch_sum/batch_sum_sq read the
  // accumulators (size nelems*ngroups); num_batches counts end_batch calls.
  virtual void end_batch() = 0;
  virtual std::vector<double> batch_sum() const = 0;
  virtual std::vector<double> batch_sum_sq() const = 0;
  virtual int64_t num_batches() const = 0;

  // Overwrite the flux tally (used by the distributed driver to install the
  // all-reduced global tally on rank 0 before writing).
  virtual void set_flux(const double *flux, int64_t nelems) = 0;

  // Restore particle state (checkpoint/resume support -- the reference has
  // none; a crash there loses the whole batch).
  virtual void set_particle_state(const double *pos, const int32_t *elem,
                                  const uint8_t *escaped, int64_t n) = 0;

  // Block until all queued device work is done (no-op on CPU).
  virtual void synchronize() {}

  // Device-resident mesh views (GPU engines only): lets sibling device
  // code (the partitioned engine's localization kernels) reuse the
  // engine's uploaded planes/grid instead of duplicating them in HBM.
  struct DeviceMeshView {
    const Plane *planes = nullptr; // nelems*4
    const int32_t *nbr = nullptr;  // nelems*4
    GridView grid{};
    // the engine's compute stream (hipStream_t; void* keeps this header
    // HIP-free): sibling kernels launched here order naturally against
    // walk_raw_device without device-wide fences
    void *stream = nullptr;
  };
  virtual bool device_mesh(DeviceMeshView *out) const {
    (void)out;
    return false;
  }

  int max_steps = 0; // 0 = auto (set by implementations from mesh size)
  int ngroups = 1;   // energy groups
  int nscores = 1;   // tally scores (flux is [nscores x ngroups x nelems])

  bool reflective = false; // specular-reflect at boundaries (else vacuum)

  // fp32-traversal fast path (walk.h walk_advance32): candidate exit-face
  // decisions in fp32, crossings/tallies in fp64.  Controlled by
  // PUMITALLY_WALK=fp32|fp64; both engines honor it so CPU remains the
  // bitwise oracle for the GPU in either mode.
  bool walk_fp32 = false;
};

inline bool default_walk_fp32() {
  const char *s = getenv("PUMITALLY_WALK");
  return s && std::string(s) == "fp32";
}

// Localization tolerance (accepted signed distance below a face plane),
// relative to the mesh bounding-box diagonal.  The reference hardcodes its
// geometric tolerance (1e-8 at PumiTallyImpl.cpp:51); here it is runtime
// configurable (SURVEY.md section 5 flags the hardcoding).
// Boundary condition: vacuum (reference parity, default) or specular
// reflective on every boundary face (symmetry-plane models).  Env:
// PUMITALLY_BC=vacuum|reflective.
inline bool default_reflective() {
  const char *s = getenv("PUMITALLY_BC");
  return s && std::string(s) == "reflective";
}

inline double loc_tol_rel() {
  const char *s = getenv("PUMITALLY_LOC_TOL");
  return s ? atof(s) : 1e-10;
}

std::unique_ptr<Engine> make_cpu_engine(Mesh mesh, int64_t num_particles,
                                        int ngroups = 1, int nscores = 1);

// Returns nullptr when no HIP device is available.
std::unique_ptr<Engine> make_gpu_engine(Mesh mesh, int64_t num_particles,
                                        int device, int ngroups = 1,
                                        int nscores = 1);

// Normalized flux = flux / element volume (volume-only, matching the
// reference implementation rather than its docstring:
// PumiTallyImpl.cpp:382-409, TODO at :372 never implemented).
std::vector<double> normalize_flux(const Mesh &m, const std::vector<double> &flux);

void write_tally_vtk(const std::string &filename, const Mesh &m,
                     const std::vector<double> &flux);

int default_max_steps(const Mesh &m);

} // namespace pumitally
