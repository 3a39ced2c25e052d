// Power-user C++ example: the Engine layer behind the PumiTally facade
// (installed as include/pumitally/{engine,mesh,geom,walk}.h).  Gives host
// codes direct access to the capabilities the 4-call facade doesn't
// expose: device-resident moves, energy groups, batch statistics,
// checkpoint-grade state access.
#include "../csrc/core/engine.h"

#include <cstdio>
#include <random>
#include <vector>

using namespace pumitally;

int main() {
  Mesh mesh = build_box(8, 8, 8, 1.0, 1.0, 1.0);

  const int64_t n = 20000;
  const int ngroups = 2;
  auto engine = make_gpu_engine(mesh, n, /*device=*/0, ngroups);
  if (!engine) {
    printf("[engine-example] no GPU; using the CPU engine\n");
    engine = make_cpu_engine(mesh, n, ngroups);
  }

  std::mt19937_64 rng(7);
  std::uniform_real_distribution<double> u(0.05, 0.95);
  std::vector<double> pos(n * 3), dest(n * 3), w(n, 1.0);
  std::vector<int8_t> fly(n, 1);
  std::vector<uint16_t> grp(n);
  for (int64_t i = 0; i < n; ++i) {
    for (int k = 0; k < 3; ++k) {
      pos[i * 3 + k] = u(rng);
      dest[i * 3 + k] = u(rng);
    }
    grp[i] = (uint16_t)(i & 1);
  }

  engine->copy_initial_position(pos.data(), n);
  for (int batch = 0; batch < 3; ++batch) {
    engine->move(pos.data(), dest.data(), fly.data(), w.data(), n, grp.data());
    engine->end_batch();
    std::swap(pos, dest); // continue from where the particles stopped
  }
  engine->synchronize();

  const auto sum = engine->batch_sum();
  double g0 = 0, g1 = 0;
  for (int64_t e = 0; e < mesh.nelems; ++e) {
    g0 += sum[e];
    g1 += sum[mesh.nelems + e];
  }
  printf("[engine-example] 3 batches, group tallies: g0=%.4f g1=%.4f, "
         "lost=%lld\n",
         g0, g1, (long long)engine->stats().lost_particles);
  return engine->stats().lost_particles == 0 ? 0 : 1;
}
