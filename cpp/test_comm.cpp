// World-2 CPU test of the library-held comm layer (csrc/comm): forks
// itself into two ranks, runs the TCP fallback's collectives, and drives
// the PumiTally facade end-to-end so rank 0's fluxresult.vtk holds the
// all-reduced two-rank tally.  No MPI, no gloo, no Python -- this is the
// C++ consumer scenario (the reference's config-5 shape at world>1,
// where the library itself owns the communication:
// /root/reference/src/pumitally/PumiTallyImpl.cpp:238-241).
#include "PumiTally.h"

#include "../csrc/comm/comm.h"
#include "../csrc/core/engine.h"
#include "../csrc/core/mesh.h"

#include <sys/wait.h>
#include <unistd.h>

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

using namespace pumitally;

static int failures = 0;
#define CHECK(cond)                                                            \
  do {                                                                         \
    if (!(cond)) {                                                             \
      fprintf(stderr, "CHECK failed at %s:%d: %s\n", __FILE__, __LINE__,       \
              #cond);                                                          \
      failures++;                                                              \
    }                                                                          \
  } while (0)

static void run_rank(int rank) {
  setenv("RANK", std::to_string(rank).c_str(), 1);
  setenv("WORLD_SIZE", "2", 1);
  setenv("MASTER_ADDR", "127.0.0.1", 1);
  setenv("PUMITALLY_PORT", "29741", 1);
  setenv("PUMITALLY_DEVICE", "cpu", 1);

  // --- raw collectives -----------------------------------------------------
  auto comm = make_comm_from_env(/*want_gpu=*/false, 0);
  CHECK(comm && comm->world() == 2 && comm->rank() == rank);

  double v[3] = {1.0 + rank, 2.0, 3.0 * (rank + 1)};
  comm->allreduce_sum(v, 3);
  CHECK(std::fabs(v[0] - 3.0) < 1e-15); // (1+0)+(1+1)
  CHECK(std::fabs(v[1] - 4.0) < 1e-15);
  CHECK(std::fabs(v[2] - 9.0) < 1e-15); // 3+6

  double mx[2] = {rank == 0 ? 5.0 : 2.0, (double)rank};
  comm->allreduce_max(mx, 2);
  CHECK(mx[0] == 5.0 && mx[1] == 1.0);

  auto g = comm->allgather(100 + rank);
  CHECK(g.size() == 2 && g[0] == 100 && g[1] == 101);

  int64_t n64[1] = {10 + rank};
  comm->allreduce_sum(n64, 1);
  CHECK(n64[0] == 21);

  char msg[8] = {0};
  if (rank == 1) strcpy(msg, "hello");
  comm->bcast(msg, 8, 1);
  CHECK(strcmp(msg, "hello") == 0);

  // alltoallv with asymmetric counts: rank0 sends {1 to r0, 3 to r1},
  // rank1 sends {2 to r0, 0 to r1}; receives concatenate in source order.
  {
    std::vector<double> send;
    std::vector<int64_t> cnt;
    if (rank == 0) {
      send = {0.5, 10.0, 11.0, 12.0};
      cnt = {1, 3};
    } else {
      send = {20.0, 21.0};
      cnt = {2, 0};
    }
    auto got = comm->alltoallv(send.data(), cnt);
    if (rank == 0) {
      CHECK(got.size() == 3 && got[0] == 0.5 && got[1] == 20.0 &&
            got[2] == 21.0);
    } else {
      CHECK(got.size() == 3 && got[0] == 10.0 && got[2] == 12.0);
    }
  }
  comm->barrier();
  comm.reset();

  // --- facade end-to-end: per-rank particles, reduced flux on rank 0 -------
  {
    Mesh box = build_box(3, 3, 3, 1.0, 1.0, 1.0);
    const std::string mesh_path = "comm_test_mesh.osh";
    if (rank == 0) write_osh(mesh_path, box);
    auto gate = make_comm_from_env(false, 0); // mesh file ready barrier
    gate->barrier();

    setenv("PUMITALLY_OUTPUT",
           rank == 0 ? "comm_test_flux.vtk" : "comm_test_flux_r1.vtk", 1);
    const int n = 4;
    int argc = 0;
    char **argv = nullptr;
    {
      PumiTally tally(mesh_path, n, argc, argv);
      std::vector<double> pos(n * 3), dest(n * 3), w(n, 1.0);
      std::vector<int8_t> fly(n, 1);
      for (int i = 0; i < n; ++i) {
        pos[i * 3] = 0.1 + 0.02 * i + 0.4 * rank; // disjoint per rank
        pos[i * 3 + 1] = 0.30;
        pos[i * 3 + 2] = 0.55;
        dest[i * 3] = pos[i * 3] + 0.25;
        dest[i * 3 + 1] = 0.30;
        dest[i * 3 + 2] = 0.55;
      }
      tally.CopyInitialPosition(pos.data(), n * 3);
      tally.MoveToNextLocation(pos.data(), dest.data(), fly.data(), w.data(),
                               n * 3);
      tally.WriteTallyResults(); // all-reduce inside; rank 0 writes
    }
    if (rank == 0) {
      // the reduced tally conserves BOTH ranks' track length: 8 segments
      // of 0.25, all interior -> total flux (pre-normalization) == 2.0.
      // Check via a fresh single-process engine replaying both batches.
      unsetenv("RANK");
      unsetenv("WORLD_SIZE");
      auto eng = make_cpu_engine(box, 8);
      std::vector<double> p2(24), d2(24), w2(8, 1.0);
      std::vector<int8_t> f2(8, 1);
      for (int r = 0; r < 2; ++r)
        for (int i = 0; i < 4; ++i) {
          const int k = r * 4 + i;
          p2[k * 3] = 0.1 + 0.02 * i + 0.4 * r;
          p2[k * 3 + 1] = 0.30;
          p2[k * 3 + 2] = 0.55;
          d2[k * 3] = p2[k * 3] + 0.25;
          d2[k * 3 + 1] = 0.30;
          d2[k * 3 + 2] = 0.55;
        }
      eng->copy_initial_position(p2.data(), 8);
      eng->move(p2.data(), d2.data(), f2.data(), w2.data(), 8);
      double want = 0.0;
      for (double f : eng->flux()) want += f;
      CHECK(std::fabs(want - 2.0) < 1e-12);
      // parse rank 0's written VTK "flux" field (big-endian doubles) and
      // compare against the oracle's normalized flux: proves the facade
      // actually all-reduced both ranks' tallies before writing
      std::vector<double> norm = normalize_flux(box, eng->flux());
      FILE *f = fopen("comm_test_flux.vtk", "rb");
      CHECK(f != nullptr);
      if (f) {
        std::string bytes;
        char buf[4096];
        size_t k;
        while ((k = fread(buf, 1, sizeof buf, f)) > 0) bytes.append(buf, k);
        fclose(f);
        const std::string tag = "SCALARS flux double 1\nLOOKUP_TABLE default\n";
        const size_t at = bytes.find(tag);
        CHECK(at != std::string::npos);
        if (at != std::string::npos) {
          // small meshes are written ASCII (one %.17g per line)
          const char *p = bytes.data() + at + tag.size();
          char *endp = nullptr;
          double maxerr = 0.0;
          for (int64_t t = 0; t < box.nelems; ++t) {
            const double val = strtod(p, &endp);
            CHECK(endp != p);
            p = endp;
            const double err = std::fabs(val - norm[t]);
            if (err > maxerr) maxerr = err;
          }
          CHECK(maxerr < 1e-12);
        }
      }
    }
  }

  if (failures) {
    fprintf(stderr, "rank %d: %d failures\n", rank, failures);
    exit(1);
  }
  printf("rank %d OK\n", rank);
}

int main(int argc, char **argv) {
  if (argc > 1 && strcmp(argv[1], "--rank") == 0) {
    run_rank(atoi(argv[2]));
    return failures ? 1 : 0;
  }
  // parent: spawn both ranks of this same binary
  pid_t kids[2];
  for (int r = 0; r < 2; ++r) {
    kids[r] = fork();
    if (kids[r] == 0) {
      char rs[8];
      snprintf(rs, sizeof rs, "%d", r);
      execl(argv[0], argv[0], "--rank", rs, (char *)nullptr);
      _exit(127);
    }
  }
  int rc = 0;
  for (int r = 0; r < 2; ++r) {
    int st = 0;
    waitpid(kids[r], &st, 0);
    if (!WIFEXITED(st) || WEXITSTATUS(st) != 0) rc = 1;
  }
  printf(rc ? "test_comm FAILED\n" : "test_comm passed\n");
  return rc;
}
