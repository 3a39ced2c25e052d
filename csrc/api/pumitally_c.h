/* Plain-C ABI over the PumiTally facade, for host transport codes that
 * are not C++ (Fortran via iso_c_binding, C, or dlopen-style embedding).
 * Mirrors the 4-call PIMPL API (PumiTally.h) 1:1; the reference offers
 * only the C++ class (reference PumiTally.h:34-107), which closes C
 * hosts out.
 *
 * Every call returns 0 on success; on failure it returns nonzero and
 * pumitally_last_error() gives the message (thread-local).  The handle
 * honors the same environment as the C++ facade (PUMITALLY_DEVICE,
 * PUMITALLY_OUTPUT, RANK/WORLD_SIZE multi-process comm, ...).
 */
#ifndef PUMITALLY_C_H
#define PUMITALLY_C_H

#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef struct pumitally_handle pumitally_handle;

/* mesh_filename: .osh directory or Gmsh .msh (ASCII or binary). */
pumitally_handle *pumitally_create(const char *mesh_filename,
                                   int32_t num_particles);

/* positions: 3*num_particles doubles (x,y,z interleaved). */
int pumitally_copy_initial_position(pumitally_handle *h, double *positions,
                                    int32_t size);

/* Per-event-step move; size = 3*num_particles.  `flying` is consumed and
 * zeroed, exactly like the C++ facade / reference host contract. */
int pumitally_move_to_next_location(pumitally_handle *h, double *origin,
                                    double *destinations, int8_t *flying,
                                    double *weights, int32_t size);

int pumitally_write_tally_results(pumitally_handle *h);

void pumitally_destroy(pumitally_handle *h);

/* Message for the most recent failing call on this thread ("" if none). */
const char *pumitally_last_error(void);

/* Test/demo helper: generate an nx*ny*nz unit-box tet mesh at `dir`
 * (.osh directory), so C hosts can self-test without a mesh file. */
int pumitally_write_box_mesh(const char *dir, int nx, int ny, int nz);

/* Accumulated phase timings in seconds (parity with TallyTimes). */
double pumitally_initialization_time(const pumitally_handle *h);
double pumitally_tally_time(const pumitally_handle *h);
double pumitally_write_time(const pumitally_handle *h);

#ifdef __cplusplus
}
#endif

#endif /* PUMITALLY_C_H */
