// Address/UB-sanitizer harness over the CPU core (mesh + walk + engine +
// partition), compiled with plain g++ (no HIP).  Runs the golden sequence
// plus randomized fuzz through every core path.
// Build/run: tools/asan_check.sh
#include "../csrc/comm/comm.h"

#include <sys/wait.h>
#include <unistd.h>
#include "../csrc/core/engine.h"

#include <cassert>
#include <cmath>
#include <cstdio>
#include <cstring>
#include <random>
#include <vector>

using namespace pumitally;

static void golden() {
  Mesh m = build_box(1, 1, 1, 1.0, 1.0, 1.0);
  auto e = make_cpu_engine(m, 5);
  std::vector<double> init(15), dest(15);
  for (int i = 0; i < 5; ++i) {
    init[i * 3] = 0.1; init[i * 3 + 1] = 0.4; init[i * 3 + 2] = 0.5;
    dest[i * 3] = 1.2; dest[i * 3 + 1] = 0.4; dest[i * 3 + 2] = 0.5;
  }
  e->copy_initial_position(init.data(), 5);
  std::vector<int8_t> fly(5, 1);
  std::vector<double> w(5, 1.0);
  e->move(init.data(), dest.data(), fly.data(), w.data(), 5);
  auto f = e->flux();
  assert(std::fabs(f[2] - 1.5) < 1e-8 && std::fabs(f[4] - 2.5) < 1e-8);
}

static void fuzz() {
  std::mt19937_64 rng(42);
  for (int trial = 0; trial < 40; ++trial) {
    std::uniform_int_distribution<int> ci(1, 5);
    const int nx = ci(rng), ny = ci(rng), nz = ci(rng);
    Mesh m = build_box(nx, ny, nz, 1.0, 1.0, 1.0);
    if (trial % 3 == 1) {
      // exercise the periodic pairing + wrap path
      std::vector<int64_t> hi, lo;
      for (int64_t t = 0; t < m.nelems; ++t)
        for (int f = 0; f < 4; ++f) {
          if (m.nbr[t * 4 + f] != -1) continue;
          double cx = 0;
          for (int k = 0; k < 3; ++k)
            cx += m.coords[(int64_t)m.tet2vert[t * 4 + kFaceVerts[f][k]] * 3];
          cx /= 3.0;
          if (std::fabs(cx - 1.0) < 1e-12) hi.push_back(t * 4 + f);
          else if (std::fabs(cx) < 1e-12) lo.push_back(t * 4 + f);
        }
      m.set_periodic_faces(hi, lo, Vec3{-1.0, 0.0, 0.0});
    }
    const int64_t n = 1 + (int64_t)(rng() % 50);
    const int nscores = 1 + (int)(rng() % 3);
    auto e = make_cpu_engine(m, n, 1, nscores);
    std::uniform_real_distribution<double> u(0.01, 0.99);
    std::vector<double> o(n * 3), d(n * 3), w(n), resp(n * nscores);
    std::vector<int8_t> fly(n, 1);
    for (int64_t i = 0; i < n * 3; ++i) { o[i] = u(rng); d[i] = u(rng); }
    for (int64_t i = 0; i < n; ++i) w[i] = u(rng);
    for (int64_t i = 0; i < n * nscores; ++i) resp[i] = u(rng);
    e->copy_initial_position(o.data(), n);
    e->move(o.data(), d.data(), fly.data(), w.data(), n, nullptr,
            nscores > 1 ? resp.data() : nullptr);
    e->move_continue(d.data(), fly.data(), w.data(), n);
    e->end_batch();
    (void)e->batch_sum();
    (void)e->positions();
    // partition paths
    auto owners = partition_morton(m, 3);
    for (int p = 0; p < 3; ++p) {
      SubMesh sub = extract_submesh(m, owners, p, 1);
      if (sub.local.nelems == 0) continue;
      auto pe = make_cpu_engine(sub.local, 1);
      std::vector<int32_t> elem(1, 0);
      std::vector<double> op(3), oe(3);
      std::vector<int32_t> out_e(1);
      std::vector<int8_t> st(1);
      double start[3] = {sub.local.centroid(0).x, sub.local.centroid(0).y,
                         sub.local.centroid(0).z};
      double stop[3] = {u(rng), u(rng), u(rng)};
      pe->walk_raw(1, start, stop, elem.data(), w.data(), op.data(),
                   out_e.data(), st.data());
    }
  }
}

// world-2 TCP comm under the sanitizers: fork two ranks, run every
// collective (ASan/UBSan follow the forked children).
static void comm_world2() {
  setenv("PUMITALLY_PORT", "29793", 1);
  pid_t kid = fork();
  const int rank = kid == 0 ? 1 : 0;
  {
    auto comm = make_tcp_comm(rank, 2, "127.0.0.1", 29793);
    double v[4] = {1.0 * rank, 2.0, 3.0, 4.0};
    comm->allreduce_sum(v, 4);
    if (v[0] != 1.0 || v[1] != 4.0) abort();
    comm->allreduce_max(v, 4);
    int64_t c[2] = {rank, 7};
    comm->allreduce_sum(c, 2);
    if (c[0] != 1 || c[1] != 14) abort();
    auto g = comm->allgather(10 + rank);
    if (g[0] != 10 || g[1] != 11) abort();
    char buf[8] = {0};
    if (rank == 1) memcpy(buf, "ok", 3);
    comm->bcast(buf, 8, 1);
    if (buf[0] != 'o') abort();
    std::vector<double> send = rank == 0
        ? std::vector<double>{1.0, 2.0, 3.0}   // 1 to r0, 2 to r1
        : std::vector<double>{9.0};            // 1 to r0, 0 to r1
    std::vector<int64_t> cnt = rank == 0 ? std::vector<int64_t>{1, 2}
                                         : std::vector<int64_t>{1, 0};
    auto got = comm->alltoallv(send.data(), cnt);
    if (rank == 0 && (got.size() != 2 || got[0] != 1.0 || got[1] != 9.0))
      abort();
    if (rank == 1 && (got.size() != 2 || got[0] != 2.0 || got[1] != 3.0))
      abort();
    comm->barrier();
  }
  if (kid == 0) _exit(0);
  int st = 0;
  waitpid(kid, &st, 0);
  if (!WIFEXITED(st) || WEXITSTATUS(st) != 0) abort();
}

// the multithreaded move path (n >= 65536) with every tally dimension
// on: per-thread partials must span the FULL flux shape (a partial
// sized nelems*ngroups overflowed under nscores > 1 -- the bug this
// harness should have caught; see tests/test_threaded_scored_move.py)
void threaded_scored_move() {
  Mesh m = build_box(8, 8, 8, 1.0, 1.0, 1.0);
  const int64_t n = 70000;
  const int G = 2, S = 2;
  auto e = make_cpu_engine(m, n, G, S);
  std::mt19937_64 rng(17);
  std::uniform_real_distribution<double> u(0.02, 0.98);
  std::vector<double> o(n * 3), d(n * 3), w(n), resp(n * S);
  std::vector<int8_t> fly(n, 1);
  std::vector<uint16_t> grp(n);
  for (int64_t i = 0; i < n * 3; ++i) { o[i] = u(rng); d[i] = u(rng); }
  for (int64_t i = 0; i < n; ++i) { w[i] = u(rng); grp[i] = rng() % G; }
  for (int64_t i = 0; i < n * S; ++i) resp[i] = u(rng);
  e->copy_initial_position(o.data(), n);
  e->move(o.data(), d.data(), fly.data(), w.data(), n, grp.data(),
          resp.data());
}

int main() {
  golden();
  fuzz();
  threaded_scored_move();
  comm_world2();
  printf("asan_check: PASS\n");
  return 0;
}
