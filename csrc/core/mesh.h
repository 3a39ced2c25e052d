// First-party unstructured tetrahedral mesh core (CPU side).
//
// Replaces the used subset of Omega_h in the reference (PUMI-Tally,
// /root/reference/src/pumitally/PumiTallyImpl.cpp:384-399,461-528,562):
//   * mesh object with coords + tet->vert connectivity
//   * derived tet->tet face adjacency (what Omega_h/pumipic derive for the
//     particle walk)
//   * per-tet volumes (Omega_h simplex_size_from_basis)
//   * build_box test-mesh generator (Omega_h::build_box, used by the
//     reference tests test/test_pumi_tally_impl_methods.cpp:34-35)
//   * mesh file IO (.osh directory format read/write - reconstructed,
//     see osh_io.cpp - plus Gmsh .msh and legacy VTK output)
//
// Unlike the reference stack there is no Kokkos view machinery: the mesh is
// flat std::vectors on the host, uploaded once into flat HBM arrays by the
// GPU engine.  The walk consumes only `planes` (4 canonically-oriented face
// planes per tet, 128 B) and `nbr` (4 neighbor ids, 16 B): one contiguous
// 144 B record per tet, no indirection to vertex coords in the hot loop.
#pragma once

#include "geom.h"

#include <array>
#include <cstdint>
#include <string>
#include <vector>

namespace pumitally {

// Local face f of a tet is the face OPPOSITE local vertex f.
// For a positively-oriented tet (v0,v1,v2,v3) the inward-positive triangles:
//   F0=(1,3,2)  F1=(0,2,3)  F2=(0,3,1)  F3=(0,1,2)
constexpr int kFaceVerts[4][3] = {{1, 3, 2}, {0, 2, 3}, {0, 3, 1}, {0, 1, 2}};

// Uniform-grid acceleration structure for point-in-mesh localization.
// CSR lists of tets whose AABB overlaps each cell.
struct LocGrid {
  int nx = 0, ny = 0, nz = 0;
  Vec3 lo{0, 0, 0};
  Vec3 inv_h{0, 0, 0}; // 1/cell_size
  std::vector<int32_t> cell_start; // size nx*ny*nz+1
  std::vector<int32_t> cell_tets;  // CSR payload
};

struct Mesh {
  int64_t nverts = 0;
  int64_t nelems = 0;
  std::vector<double> coords;   // nverts*3, xyz interleaved
  std::vector<int32_t> tet2vert; // nelems*4, positively oriented

  // Optional per-face boundary conditions: bit set => that boundary face
  // reflects (specular); unset/absent => vacuum.  Indexed by elem*4+f;
  // only meaningful where nbr==-1.  Set via set_face_reflective BEFORE
  // engine construction (engines snapshot it).  The global
  // PUMITALLY_BC=reflective mode overrides everything to reflective.
  std::vector<uint32_t> face_bc_bits;
  void set_face_reflective(int64_t face_index);
  bool face_is_reflective(int64_t face_index) const {
    const size_t w = (size_t)(face_index >> 5);
    return w < face_bc_bits.size() &&
           ((face_bc_bits[w] >> (face_index & 31)) & 1u);
  }

  // Optional periodic boundary pairing: boundary face a (elem*4+f) maps to
  // an entry element + translation vector; a walk hitting the face
  // teleports its remaining segment by the translation and resumes there
  // (walk.h periodic_restart).  Built by set_periodic_faces(a, b, T):
  // every face in `a`, translated by T, must coincide with a face in `b`
  // (matched by face centroid to tol); pairing is installed both ways
  // (b gets -T).  Call AFTER finalize() and BEFORE engine construction.
  // Partitioned support: extract_submesh wires pairs whose partner is in
  // the submesh into the local tables and turns cross-part pairs into
  // translation-carrying handoff entries (SubMesh::foreign_shift); the
  // stateful PartitionedEngine accepts periodic only at world 1.
  std::vector<int32_t> periodic_idx;   // nelems*4, -1 = no pair; empty = none
  std::vector<int32_t> periodic_elem;  // pair entry -> entry element
  std::vector<double> periodic_shift;  // pair entry -> translation (x,y,z)
  void set_periodic_faces(const std::vector<int64_t> &faces_a,
                          const std::vector<int64_t> &faces_b,
                          Vec3 translation, double tol = -1.0);
  bool has_periodic() const { return !periodic_elem.empty(); }

  // Derived (built by finalize()):
  std::vector<int32_t> nbr;     // nelems*4: neighbor tet across face f, -1 = boundary
  std::vector<Plane> planes;    // nelems*4: inward-positive unit-normal face planes
  std::vector<Plane32> planes32; // fp32 copies (64 B/tet) for the traversal fast path
  std::vector<double> volumes;  // nelems
  Vec3 bbox_lo{0, 0, 0}, bbox_hi{0, 0, 0};
  LocGrid grid;

  // Build adjacency, planes, volumes, bbox and the localization grid.
  // Reorients negatively-oriented tets in place (swaps verts 2,3).
  void finalize();

  Vec3 vert(int32_t v) const { return {coords[v * 3], coords[v * 3 + 1], coords[v * 3 + 2]}; }
  Vec3 centroid(int32_t t) const;

  // Point-in-tet test used by localization (CPU path). tol: accepted signed
  // distance below a face plane (>=0 means strictly inside).
  bool contains(int32_t t, Vec3 p, double tol) const;
  // Locate the tet containing p, or -1. CPU reference implementation of the
  // GPU localization kernel.
  // used_loose (optional): set to true when only the tol*1e4 retry pass
  // succeeded (see EngineStats::loose_localizations).
  int32_t locate(Vec3 p, double tol, bool *used_loose = nullptr) const;
};

// Analytic box mesh generator: divisions (nx,ny,nz) over extents (lx,ly,lz).
// Each grid cell is cut into 6 tets around its main diagonal using the same
// element ordering as Omega_h::build_box for the unit cube, which the
// reference tests pin (element ids 2,3,4 along the x-ray, centroid of
// element 0 at (0.5,0.75,0.25); /root/reference/test/
// test_pumi_tally_impl_methods.cpp:83,152-159,221-282).
Mesh build_box(int nx, int ny, int nz, double lx, double ly, double lz);

Mesh mesh_from_arrays(int64_t nverts, const double *coords, int64_t nelems,
                      const int32_t *tets);

// Domain decomposition (partition.cpp).  Element-ownership partition of a
// finalized mesh: the MI355X-native replacement for pumipic::Mesh picparts
// (reference PumiTallyImpl.cpp:530-539, degenerate there: all owners rank 0).
struct SubMesh {
  Mesh local;                        // owned elements, locally-renumbered verts
  std::vector<int64_t> elem_l2g;     // local elem -> global elem
  std::vector<int64_t> foreign_gid;  // handoff table k -> global elem id
  std::vector<int32_t> foreign_owner; // handoff table k -> owning part
  // handoff table k -> periodic translation to apply to the particle's
  // position AND destination when shipping through this face (zero for
  // plain partition cuts; nonzero when the global face is a periodic
  // pair whose partner element lives on another part).  Local periodic
  // pairs (partner inside this submesh) are wired straight into
  // local.periodic_* and never reach the handoff table.
  std::vector<double> foreign_shift; // k*3
};

// Balanced spatial partition: elements sorted by Morton key of centroid,
// split into nparts equal chunks.  Returns per-element owner ids.
// weights (nullable, nelems doubles): per-element work estimates; when
// given, chunks equalize summed weight instead of element count (dynamic
// load balance for partitioned tallies -- feed the previous batch's raw
// flux back in and re-extract submeshes).
std::vector<int32_t> partition_morton(const Mesh &m, int nparts,
                                      const double *weights = nullptr);

// Extract part `part`'s submesh.  Local faces whose global neighbor is
// not in the submesh get nbr = -(2+k) with foreign_gid[k]/foreign_owner[k]
// describing the remote element; the walk stops there (walk.h).  Local
// vertex ids ascend with global ids, so canonical face planes are
// bitwise-identical to the full mesh's -- cross-rank walks tile segments
// exactly like a single-mesh walk.
//
// ghost_rings: include that many rings of face-neighbor elements around
// the owned set.  Particles keep walking through ghost elements (tallying
// locally -- ghost-element tallies are summed by the flux all-reduce) and
// hand off only when leaving the owned+ghost region, cutting the number
// of exchange rounds for cut-hugging tracks.
SubMesh extract_submesh(const Mesh &m, const std::vector<int32_t> &owners,
                        int part, int ghost_rings = 0);

// Best-effort Omega_h binary stream reader (osh_omegah.cpp): probed
// layout + orientation-free reconstruction, refuses anything it cannot
// prove consistent.  Called by read_osh when the Omega_h magic is seen.
Mesh read_osh_omegah_stream(const std::string &stream_path);

// IO (implemented in mesh_io.cpp / osh_io.cpp)
Mesh read_gmsh(const std::string &path);             // Gmsh .msh v2.2/v4.1 ASCII
Mesh read_mesh(const std::string &path);             // dispatch on extension
// binary: 1 = legacy binary (big-endian), 0 = ASCII, -1 = auto (binary for
// meshes over 200k elements -- ~10x smaller/faster finalization).
void write_vtk(const std::string &path, const Mesh &m,
               const std::vector<std::pair<std::string, std::vector<double>>> &cell_data,
               int binary = -1);
Mesh read_osh(const std::string &dir);               // .osh directory
void write_osh(const std::string &dir, const Mesh &m);

} // namespace pumitally
