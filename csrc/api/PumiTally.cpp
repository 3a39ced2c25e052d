#include "PumiTally.h"

#include "../comm/comm.h"
#include "../core/engine.h"

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <string>

namespace pumitally {

namespace {
double now_s() {
  using clk = std::chrono::steady_clock;
  return std::chrono::duration<double>(clk::now().time_since_epoch()).count();
}
} // namespace

// Accumulated wall-clock phase timers; equivalent of the reference
// TallyTimes (PumiTallyImpl.h:18-27) with working device fences.
struct TallyTimes {
  double initialization_time = 0.0;
  double total_time_to_tally = 0.0;
  double vtk_file_write_time = 0.0;
  void print() const {
    printf("\n");
    printf("[TIME] Initialization time     : %f seconds\n", initialization_time);
    printf("[TIME] Total time to tally     : %f seconds\n", total_time_to_tally);
    printf("[TIME] VTK file write time     : %f seconds\n", vtk_file_write_time);
    printf("[TIME] Total PUMI-Tally time   : %f seconds\n",
           initialization_time + total_time_to_tally + vtk_file_write_time);
  }
};

struct PumiTallyImpl {
  std::unique_ptr<Engine> engine;
  std::unique_ptr<Comm> comm; // null in single-process runs
  int32_t num_particles = 0;
  TallyTimes times;
  std::string output = "fluxresult.vtk";
};

PumiTally::PumiTally(const std::string &mesh_filename, int32_t num_particles,
                     int &argc, char **&argv) {
  (void)argc;
  (void)argv;
  pimpl_ = std::make_unique<PumiTallyImpl>();
  pimpl_->num_particles = num_particles;
  const double t0 = now_s();
  Mesh mesh = read_mesh(mesh_filename);

  // Library-held multi-process support (the reference's pumipic::Library
  // holds the MPI world comm, PumiTallyImpl.cpp:238-241): rank/world come
  // from torchrun-compatible env; each rank walks its own particle batch
  // on its own GPU (LOCAL_RANK) against the replicated mesh, and
  // WriteTallyResults all-reduces the flux over RCCL before rank 0 writes.
  const EnvComm env = comm_env();
  const char *dev = getenv("PUMITALLY_DEVICE");
  std::unique_ptr<Engine> eng;
  if (!dev || std::string(dev) != "cpu") {
    const int ordinal = dev ? atoi(dev) : env.local_rank;
    eng = make_gpu_engine(mesh, num_particles, ordinal,
                          /*ngroups=*/1, /*nscores=*/1);
  }
  const bool on_gpu = eng != nullptr;
  if (!eng) eng = make_cpu_engine(std::move(mesh), num_particles);
  pimpl_->engine = std::move(eng);
  pimpl_->comm = make_comm_from_env(on_gpu, env.local_rank);
  if (!pimpl_->comm || pimpl_->comm->rank() == 0)
    printf("[INFO] pumitally loaded mesh %s with %lld elements%s\n",
           mesh_filename.c_str(), (long long)pimpl_->engine->mesh().nelems,
           pimpl_->comm
               ? (" (world " + std::to_string(pimpl_->comm->world()) + ")")
                     .c_str()
               : "");
  if (const char *out = getenv("PUMITALLY_OUTPUT")) pimpl_->output = out;
  pimpl_->engine->synchronize();
  pimpl_->times.initialization_time += now_s() - t0;
}

void PumiTally::CopyInitialPosition(double *init_particle_positions,
                                    std::int32_t size) const {
  const double t0 = now_s();
  if (size != pimpl_->num_particles * 3)
    throw std::runtime_error("CopyInitialPosition: size must be 3*num_particles");
  pimpl_->engine->copy_initial_position(init_particle_positions,
                                        pimpl_->num_particles);
  pimpl_->engine->synchronize();
  pimpl_->times.initialization_time += now_s() - t0;
}

void PumiTally::MoveToNextLocation(double *particle_origin,
                                   double *particle_destinations,
                                   int8_t *flying, double *weights,
                                   int32_t size) const {
  const double t0 = now_s();
  if (size != pimpl_->num_particles * 3)
    throw std::runtime_error("MoveToNextLocation: size must be 3*num_particles");
  pimpl_->engine->move(particle_origin, particle_destinations, flying, weights,
                       pimpl_->num_particles);
  // Parity with the reference host contract: the flying array is consumed
  // and zeroed after upload (PumiTallyImpl.cpp:169-172).
  for (int32_t i = 0; i < pimpl_->num_particles; ++i) flying[i] = 0;
  pimpl_->engine->synchronize();
  pimpl_->times.total_time_to_tally += now_s() - t0;
}

void PumiTally::WriteTallyResults() const {
  const double t0 = now_s();
  pimpl_->engine->synchronize();
  std::vector<double> flux = pimpl_->engine->flux();
  if (pimpl_->comm) {
    // the one collective of the whole run: sum the per-rank tallies
    // (RCCL over xGMI on GPU engines, TCP fallback on CPU)
    pimpl_->comm->allreduce_sum(flux.data(), (int64_t)flux.size());
  }
  if (!pimpl_->comm || pimpl_->comm->rank() == 0)
    write_tally_vtk(pimpl_->output, pimpl_->engine->mesh(), flux);
  if (pimpl_->comm) pimpl_->comm->barrier();
  pimpl_->times.vtk_file_write_time += now_s() - t0;
  if (!pimpl_->comm || pimpl_->comm->rank() == 0) pimpl_->times.print();
}

double PumiTally::InitializationTime() const {
  return pimpl_->times.initialization_time;
}
double PumiTally::TallyTime() const { return pimpl_->times.total_time_to_tally; }
double PumiTally::WriteTime() const { return pimpl_->times.vtk_file_write_time; }

PumiTally::~PumiTally() = default;

} // namespace pumitally
