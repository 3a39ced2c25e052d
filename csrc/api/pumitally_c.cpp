// C-ABI implementation (see pumitally_c.h).
#include "pumitally_c.h"

#include "PumiTally.h"

#include "../core/mesh.h"

#include <exception>
#include <memory>
#include <string>

namespace {
thread_local std::string g_last_error;

template <class F> int guarded(F &&f) {
  try {
    f();
    g_last_error.clear();
    return 0;
  } catch (const std::exception &e) {
    g_last_error = e.what();
    return 1;
  } catch (...) {
    g_last_error = "unknown error";
    return 1;
  }
}
} // namespace

struct pumitally_handle {
  std::unique_ptr<pumitally::PumiTally> tally;
};

extern "C" {

pumitally_handle *pumitally_create(const char *mesh_filename,
                                   int32_t num_particles) {
  pumitally_handle *h = nullptr;
  const int rc = guarded([&] {
    int argc = 0;
    char **argv = nullptr;
    auto t = std::make_unique<pumitally::PumiTally>(
        std::string(mesh_filename ? mesh_filename : ""), num_particles, argc,
        argv);
    h = new pumitally_handle{std::move(t)};
  });
  return rc == 0 ? h : nullptr;
}

int pumitally_copy_initial_position(pumitally_handle *h, double *positions,
                                    int32_t size) {
  if (!h) return (g_last_error = "null handle", 1);
  return guarded([&] { h->tally->CopyInitialPosition(positions, size); });
}

int pumitally_move_to_next_location(pumitally_handle *h, double *origin,
                                    double *destinations, int8_t *flying,
                                    double *weights, int32_t size) {
  if (!h) return (g_last_error = "null handle", 1);
  return guarded([&] {
    h->tally->MoveToNextLocation(origin, destinations, flying, weights, size);
  });
}

int pumitally_write_tally_results(pumitally_handle *h) {
  if (!h) return (g_last_error = "null handle", 1);
  return guarded([&] { h->tally->WriteTallyResults(); });
}

void pumitally_destroy(pumitally_handle *h) { delete h; }

const char *pumitally_last_error(void) { return g_last_error.c_str(); }

int pumitally_write_box_mesh(const char *dir, int nx, int ny, int nz) {
  return guarded([&] {
    pumitally::Mesh m = pumitally::build_box(nx, ny, nz, 1.0, 1.0, 1.0);
    pumitally::write_osh(dir ? dir : "box.osh", m);
  });
}

double pumitally_initialization_time(const pumitally_handle *h) {
  return h ? h->tally->InitializationTime() : 0.0;
}
double pumitally_tally_time(const pumitally_handle *h) {
  return h ? h->tally->TallyTime() : 0.0;
}
double pumitally_write_time(const pumitally_handle *h) {
  return h ? h->tally->WriteTime() : 0.0;
}

} // extern "C"
