// C++ golden-value test of the PumiTally facade (no test framework needed).
// Re-encodes the reference integration test values
// (/root/reference/test/test_pumi_tally_impl_methods.cpp) against the
// public 4-call API: 5 particles, 6-tet unit cube, exact track lengths.
#include "PumiTally.h"

#include "../csrc/core/mesh.h"

#include <cmath>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

static int failures = 0;
#define CHECK(cond)                                                            \
  do {                                                                         \
    if (!(cond)) {                                                             \
      fprintf(stderr, "FAIL %s:%d: %s\n", __FILE__, __LINE__, #cond);          \
      failures++;                                                              \
    }                                                                          \
  } while (0)

static bool close8(double a, double b) { return std::fabs(a - b) < 1e-8; }

int main(int argc, char **argv) {
  setenv("PUMITALLY_DEVICE", getenv("PUMITALLY_TEST_DEVICE")
                                 ? getenv("PUMITALLY_TEST_DEVICE")
                                 : "cpu", 0);
  setenv("PUMITALLY_OUTPUT", "cpp_fluxresult.vtk", 1);

  // write the 6-tet unit cube fixture
  pumitally::Mesh box = pumitally::build_box(1, 1, 1, 1.0, 1.0, 1.0);
  pumitally::write_osh("cpp_test_mesh.osh", box);

  const int n = 5;
  pumitally::PumiTally tally("cpp_test_mesh.osh", n, argc, argv);

  std::vector<double> init(n * 3);
  for (int i = 0; i < n; ++i) {
    init[i * 3] = 0.1;
    init[i * 3 + 1] = 0.4;
    init[i * 3 + 2] = 0.5;
  }
  tally.CopyInitialPosition(init.data(), n * 3);

  std::vector<double> dest(n * 3);
  for (int i = 0; i < n; ++i) {
    dest[i * 3] = 1.2;
    dest[i * 3 + 1] = 0.4;
    dest[i * 3 + 2] = 0.5;
  }
  std::vector<int8_t> flying(n, 1);
  std::vector<double> weights(n, 1.0);
  tally.MoveToNextLocation(init.data(), dest.data(), flying.data(),
                           weights.data(), n * 3);
  for (int i = 0; i < n; ++i) CHECK(flying[i] == 0); // consumed & zeroed

  tally.WriteTallyResults();

  // parse the VTK back and check normalized flux (flux/volume, volume=1/6)
  FILE *f = fopen("cpp_fluxresult.vtk", "r");
  CHECK(f != nullptr);
  if (f) {
    char line[512];
    std::vector<double> flux;
    bool in_flux = false;
    while (fgets(line, sizeof line, f)) {
      if (strstr(line, "SCALARS flux")) {
        fgets(line, sizeof line, f); // LOOKUP_TABLE
        for (int e = 0; e < 6; ++e) {
          fgets(line, sizeof line, f);
          flux.push_back(atof(line));
        }
        in_flux = true;
        break;
      }
    }
    CHECK(in_flux);
    if (flux.size() == 6) {
      CHECK(close8(flux[0], 0.0));
      CHECK(close8(flux[1], 0.0));
      CHECK(close8(flux[2], 0.3 * n * 6)); // 6 = 1/volume
      CHECK(close8(flux[3], 0.1 * n * 6));
      CHECK(close8(flux[4], 0.5 * n * 6));
      CHECK(close8(flux[5], 0.0));
    }
    fclose(f);
  }

  if (failures == 0) printf("C++ API golden test: PASS\n");
  return failures ? 1 : 0;
}
