// Standalone host-application driver: the OpenMC-integration shape
// (reference README.md:82-134) without OpenMC -- a CPU "transport loop"
// driving the tally engine through the public 4-call API with synthetic
// straight-line histories.  This is the C++ twin of bench.py.
//
// Usage: pumitally_driver [mesh.osh|mesh.msh] [num_particles] [steps]
//        pumitally_driver --ohMesh mesh.osh [num_particles] [steps]
//        (no mesh argument: generates a 1M-tet box)
// `--ohMesh` mirrors the reference host-app CLI (`openmc --ohMesh mesh.osh`,
// reference README.md:131).
#include "PumiTally.h"

#include "../csrc/core/mesh.h"

#include <chrono>
#include <thread>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <random>
#include <string>
#include <vector>

int main(int argc, char **argv) {
  // accept the reference's `--ohMesh <path>` spelling transparently
  int a = 1;
  std::string mesh_path;
  if (a < argc && std::string(argv[a]) == "--ohMesh") {
    if (a + 1 >= argc) {
      fprintf(stderr, "--ohMesh requires a mesh path\n");
      return 2;
    }
    mesh_path = argv[a + 1];
    a += 2;
  } else if (a < argc) {
    mesh_path = argv[a++];
  }
  const int n = a < argc ? atoi(argv[a++]) : 1'000'000;
  const int steps = a < argc ? atoi(argv[a++]) : 10;

  // Multi-process launch (library-held comm, csrc/comm): RANK/WORLD_SIZE
  // in the env make each process walk its own particle stream on its own
  // GPU; the facade all-reduces the flux at WriteTallyResults and rank 0
  // writes the single fluxresult.vtk.  e.g.
  //   for r in 0..7: RANK=$r WORLD_SIZE=8 pumitally_driver --ohMesh m.osh &
  const char *rank_env = getenv("RANK");
  const int rank = rank_env ? atoi(rank_env) : 0;

  if (mesh_path.empty()) {
    if (rank == 0) {
      printf("[driver] generating 1M-tet box mesh...\n");
      pumitally::Mesh box = pumitally::build_box(55, 55, 55, 1.0, 1.0, 1.0);
      pumitally::write_osh("driver_mesh.osh", box);
    }
    mesh_path = "driver_mesh.osh";
  }
  if (rank > 0) {
    // wait for rank 0's generated mesh to appear (no comm exists yet)
    for (int w = 0; w < 600; ++w) {
      FILE *probe = fopen((mesh_path + "/nparts").c_str(), "rb");
      if (!probe) probe = fopen(mesh_path.c_str(), "rb");
      if (probe) {
        fclose(probe);
        break;
      }
      std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
  }

  pumitally::PumiTally tally(mesh_path, n, argc, argv);

  std::mt19937_64 rng(12345 + 7919ull * rank); // disjoint per-rank streams
  std::uniform_real_distribution<double> upos(0.01, 0.99);
  std::uniform_real_distribution<double> u01(0.0, 1.0);

  std::vector<double> pos(n * 3), dest(n * 3), weights(n);
  std::vector<int8_t> flying(n);
  for (int i = 0; i < n; ++i) {
    pos[i * 3] = upos(rng);
    pos[i * 3 + 1] = upos(rng);
    pos[i * 3 + 2] = upos(rng);
    weights[i] = 0.25 + 0.75 * u01(rng);
  }
  tally.CopyInitialPosition(pos.data(), n * 3);

  const double seg = 0.05;
  auto t0 = std::chrono::steady_clock::now();
  for (int s = 0; s < steps; ++s) {
    for (int i = 0; i < n; ++i) {
      const double mu = 2.0 * u01(rng) - 1.0;
      const double phi = 6.283185307179586 * u01(rng);
      const double st = std::sqrt(1.0 - mu * mu);
      double d[3] = {seg * st * std::cos(phi), seg * st * std::sin(phi), seg * mu};
      for (int k = 0; k < 3; ++k) {
        double x = pos[i * 3 + k] + d[k];
        x = std::fabs(x);
        if (x > 1.0) x = 2.0 - x;
        dest[i * 3 + k] = x;
      }
      flying[i] = 1;
    }
    tally.MoveToNextLocation(pos.data(), dest.data(), flying.data(),
                             weights.data(), n * 3);
    std::memcpy(pos.data(), dest.data(), sizeof(double) * n * 3);
  }
  const double dt =
      std::chrono::duration<double>(std::chrono::steady_clock::now() - t0)
          .count();
  printf("[driver] %d particles x %d steps in %.3f s  ->  %.2fM particle-steps/s\n",
         n, steps, dt, (double)n * steps / dt / 1e6);

  tally.WriteTallyResults();
  return 0;
}
