// Public C++ interface of the MI355X-native PUMI-Tally re-implementation.
//
// API-surface parity with the reference
// (/root/reference/src/pumitally/PumiTally.h:34-107): the same four calls
// with the same signatures and the same PIMPL pattern, so a physics code
// integrated against the reference (e.g. the OpenMC fork's
// --ohMesh path, reference README.md:82-134) recompiles against this
// header unchanged.  Only builtin types appear here; all mesh/engine state
// lives behind the pointer.
//
// Differences from the reference (documented, intentional):
//   * argc/argv are accepted for signature parity but ignored: there is no
//     Kokkos and no implicit MPI here.  Device selection: the engine runs
//     on HIP device 0 when one is present, else on the CPU.  Set
//     PUMITALLY_DEVICE=cpu|<ordinal> to override.
//   * Timing (TallyTimes equivalent) uses real device synchronization; the
//     reference's phase fences were dead code due to a macro-name mismatch
//     (PUMI_MEASURE_TIME vs PUMITALLY_MEASURE_TIME, CMakeLists.txt:68-75).
//   * Output is a single file (default "fluxresult.vtk", override with
//     PUMITALLY_OUTPUT; a path ending in .vtu writes modern XML
//     UnstructuredGrid instead of legacy VTK).
#ifndef PUMITALLY_PUMITALLY_H
#define PUMITALLY_PUMITALLY_H

#include <cstdint>
#include <memory>
#include <string>

namespace pumitally {

struct PumiTallyImpl;

class PumiTally {
public:
  // Read the mesh (.osh directory or Gmsh .msh) and initialize the particle
  // arrays; all particles start at the centroid of element 0.
  PumiTally(const std::string &mesh_filename, int32_t num_particles, int &argc,
            char **&argv);

  // One-time localization of the sampled source positions (flattened
  // x1,y1,z1,x2,...; size = 3*num_particles).  No tallying.
  void CopyInitialPosition(double *init_particle_positions,
                           std::int32_t size) const;

  // Per-transport-step move: relocate flying particles to particle_origin
  // (untallied), then track them to particle_destinations accumulating
  // track_length*weight per element crossed.  flying: 1=moving 0=stopped.
  // size = 3*num_particles.
  void MoveToNextLocation(double *particle_origin,
                          double *particle_destinations, int8_t *flying,
                          double *weights, int32_t size) const;

  // Normalize the tally by element volume and write the VTK file; prints
  // accumulated phase timings.
  void WriteTallyResults() const;

  ~PumiTally();

  // Accumulated wall-clock phase timings (seconds): initialization (mesh
  // load + localization), tally (move calls, device-synchronized), vtk
  // write.  The reference collects the same three but its device fences
  // were dead code (macro-name mismatch); these are accurate.
  double InitializationTime() const;
  double TallyTime() const;
  double WriteTime() const;

private:
  std::unique_ptr<PumiTallyImpl> pimpl_;
};

} // namespace pumitally

#endif // PUMITALLY_PUMITALLY_H
