#include "mesh.h"

#include <algorithm>
#include <cmath>
#include <stdexcept>
#include <unordered_map>

namespace pumitally {

void Mesh::set_face_reflective(int64_t face_index) {
  const size_t need = (size_t)((nelems * 4 + 31) / 32);
  if (face_bc_bits.size() < need) face_bc_bits.resize(need, 0);
  face_bc_bits[face_index >> 5] |= 1u << (face_index & 31);
}

Vec3 Mesh::centroid(int32_t t) const {
  Vec3 c{0, 0, 0};
  for (int k = 0; k < 4; ++k) c = c + vert(tet2vert[t * 4 + k]);
  return 0.25 * c;
}

namespace {
inline Vec3 face_centroid(const Mesh &m, int64_t fidx) {
  const int32_t t = (int32_t)(fidx / 4);
  const int f = (int)(fidx % 4);
  Vec3 c{0, 0, 0};
  for (int k = 0; k < 3; ++k)
    c = c + m.vert(m.tet2vert[(int64_t)t * 4 + kFaceVerts[f][k]]);
  return (1.0 / 3.0) * c;
}
} // namespace

void Mesh::set_periodic_faces(const std::vector<int64_t> &faces_a,
                              const std::vector<int64_t> &faces_b,
                              Vec3 translation, double tol) {
  if (nbr.empty())
    throw std::runtime_error("set_periodic_faces: call finalize() first");
  if (faces_a.size() != faces_b.size())
    throw std::runtime_error("set_periodic_faces: face lists differ in size");
  if (tol <= 0.0) tol = 1e-8 * norm(bbox_hi - bbox_lo);
  for (int64_t fidx : faces_a)
    if (fidx < 0 || fidx >= nelems * 4 || nbr[fidx] != -1)
      throw std::runtime_error("set_periodic_faces: face in A is not a "
                               "boundary face: " + std::to_string(fidx));
  for (int64_t fidx : faces_b)
    if (fidx < 0 || fidx >= nelems * 4 || nbr[fidx] != -1)
      throw std::runtime_error("set_periodic_faces: face in B is not a "
                               "boundary face: " + std::to_string(fidx));
  // Match face a + T to face b by centroid: sort B by x and scan a window.
  struct Entry { double x; Vec3 c; int64_t fidx; };
  std::vector<Entry> bs;
  bs.reserve(faces_b.size());
  for (int64_t fidx : faces_b) {
    const Vec3 c = face_centroid(*this, fidx);
    bs.push_back({c.x, c, fidx});
  }
  std::sort(bs.begin(), bs.end(),
            [](const Entry &l, const Entry &r) { return l.x < r.x; });
  if (periodic_idx.empty()) periodic_idx.assign(nelems * 4, -1);
  std::vector<char> taken(bs.size(), 0);
  for (int64_t fa : faces_a) {
    const Vec3 want = face_centroid(*this, fa) + translation;
    auto lo = std::lower_bound(
        bs.begin(), bs.end(), want.x - tol,
        [](const Entry &e, double v) { return e.x < v; });
    int64_t fb = -1;
    for (auto it = lo; it != bs.end() && it->x <= want.x + tol; ++it) {
      if (taken[it - bs.begin()]) continue;
      if (norm(it->c - want) <= tol) {
        fb = it->fidx;
        taken[it - bs.begin()] = 1;
        break;
      }
    }
    if (fb < 0)
      throw std::runtime_error(
          "set_periodic_faces: no face in B matches face " +
          std::to_string(fa) + " translated by T (geometry mismatch?)");
    // a -> elem(b) with +T; b -> elem(a) with -T
    const int32_t ka = (int32_t)periodic_elem.size();
    periodic_elem.push_back((int32_t)(fb / 4));
    periodic_shift.insert(periodic_shift.end(),
                          {translation.x, translation.y, translation.z});
    periodic_idx[fa] = ka;
    const int32_t kb = (int32_t)periodic_elem.size();
    periodic_elem.push_back((int32_t)(fa / 4));
    periodic_shift.insert(periodic_shift.end(),
                          {-translation.x, -translation.y, -translation.z});
    periodic_idx[fb] = kb;
  }
}

bool Mesh::contains(int32_t t, Vec3 p, double tol) const {
  for (int f = 0; f < 4; ++f)
    if (plane_eval(planes[t * 4 + f], p) < -tol) return false;
  return true;
}

namespace {
// Key for matching tet faces: sorted vertex triple.
struct FaceKey {
  int32_t a, b, c;
  bool operator==(const FaceKey &o) const { return a == o.a && b == o.b && c == o.c; }
};
struct FaceKeyHash {
  size_t operator()(const FaceKey &k) const {
    size_t h = (size_t)k.a * 0x9e3779b97f4a7c15ull;
    h ^= (size_t)k.b + 0x9e3779b97f4a7c15ull + (h << 6) + (h >> 2);
    h ^= (size_t)k.c + 0x9e3779b97f4a7c15ull + (h << 6) + (h >> 2);
    return h;
  }
};
} // namespace

void Mesh::finalize() {
  // 1. Ensure positive orientation (Omega_h meshes are positively oriented by
  //    convention; generated/imported meshes may not be).
  for (int64_t t = 0; t < nelems; ++t) {
    int32_t *tv = &tet2vert[t * 4];
    const double v6 = 6.0 * signed_volume(vert(tv[0]), vert(tv[1]), vert(tv[2]), vert(tv[3]));
    if (v6 < 0) std::swap(tv[2], tv[3]);
  }

  // 2. Face adjacency by hashing sorted vertex triples.
  nbr.assign(nelems * 4, -1);
  {
    std::unordered_map<FaceKey, int64_t, FaceKeyHash> open; // key -> t*4+f
    open.reserve(nelems * 2);
    for (int64_t t = 0; t < nelems; ++t) {
      for (int f = 0; f < 4; ++f) {
        int32_t v[3] = {tet2vert[t * 4 + kFaceVerts[f][0]],
                        tet2vert[t * 4 + kFaceVerts[f][1]],
                        tet2vert[t * 4 + kFaceVerts[f][2]]};
        std::sort(v, v + 3);
        FaceKey key{v[0], v[1], v[2]};
        auto it = open.find(key);
        if (it == open.end()) {
          open.emplace(key, t * 4 + f);
        } else if (it->second < 0) {
          throw std::runtime_error(
              "non-manifold mesh: face shared by more than two tets (tet " +
              std::to_string(t) + ")");
        } else {
          const int64_t other = it->second;
          nbr[t * 4 + f] = (int32_t)(other / 4);
          nbr[other] = (int32_t)t;
          it->second = -1; // paired; a third occurrence is non-manifold
        }
      }
    }
  }

  // 3. Face planes, canonically oriented.
  //
  // The plane of a shared face must be EXACTLY consistent (bitwise negated)
  // between its two tets so that the crossing parameter t agrees and the walk
  // never double-counts or skips a sliver at a face.  We therefore compute
  // each plane from the face's vertices in sorted-global-id order (canonical,
  // identical on both sides) and then flip the sign for whichever tet has its
  // opposite vertex on the negative side -- sign flips are exact in IEEE754.
  planes.resize(nelems * 4);
  volumes.resize(nelems);
  for (int64_t t = 0; t < nelems; ++t) {
    const Vec3 tv[4] = {vert(tet2vert[t * 4]), vert(tet2vert[t * 4 + 1]),
                        vert(tet2vert[t * 4 + 2]), vert(tet2vert[t * 4 + 3])};
    volumes[t] = signed_volume(tv[0], tv[1], tv[2], tv[3]);
    if (!(volumes[t] > 0.0))
      throw std::runtime_error("degenerate element " + std::to_string(t) +
                               " (zero or NaN volume)");
    for (int f = 0; f < 4; ++f) {
      int32_t gv[3] = {tet2vert[t * 4 + kFaceVerts[f][0]],
                       tet2vert[t * 4 + kFaceVerts[f][1]],
                       tet2vert[t * 4 + kFaceVerts[f][2]]};
      std::sort(gv, gv + 3);
      const Vec3 a = vert(gv[0]), b = vert(gv[1]), c = vert(gv[2]);
      Vec3 n = cross(b - a, c - a);
      const double len = norm(n);
      if (len == 0.0) throw std::runtime_error("degenerate face in mesh");
      n = (1.0 / len) * n;
      double off = dot(n, a);
      // opposite vertex decides the inward sign
      const Vec3 opp = tv[f];
      if (dot(n, opp) - off < 0) {
        n = {-n.x, -n.y, -n.z};
        off = -off;
      }
      planes[t * 4 + f] = Plane{n.x, n.y, n.z, off};
    }
  }

  planes32.resize(planes.size());
  for (size_t i = 0; i < planes.size(); ++i)
    planes32[i] = Plane32{(float)planes[i].nx, (float)planes[i].ny,
                          (float)planes[i].nz, (float)planes[i].c};

  // 4. bbox
  bbox_lo = {1e300, 1e300, 1e300};
  bbox_hi = {-1e300, -1e300, -1e300};
  for (int64_t v = 0; v < nverts; ++v) {
    const Vec3 p = vert((int32_t)v);
    bbox_lo.x = std::min(bbox_lo.x, p.x); bbox_hi.x = std::max(bbox_hi.x, p.x);
    bbox_lo.y = std::min(bbox_lo.y, p.y); bbox_hi.y = std::max(bbox_hi.y, p.y);
    bbox_lo.z = std::min(bbox_lo.z, p.z); bbox_hi.z = std::max(bbox_hi.z, p.z);
  }

  // 5. Localization grid: ~2 tets per cell target, padded bbox.
  {
    const double target_cells = std::max<double>(1.0, (double)nelems / 2.0);
    int n1 = (int)std::ceil(std::cbrt(target_cells));
    n1 = std::max(1, std::min(n1, 1024));
    grid.nx = grid.ny = grid.nz = n1;
    Vec3 ext = bbox_hi - bbox_lo;
    const double pad = 1e-9 * std::max({ext.x, ext.y, ext.z, 1.0});
    grid.lo = bbox_lo - Vec3{pad, pad, pad};
    Vec3 span = (bbox_hi + Vec3{pad, pad, pad}) - grid.lo;
    span.x = std::max(span.x, 1e-300); span.y = std::max(span.y, 1e-300);
    span.z = std::max(span.z, 1e-300);
    grid.inv_h = {grid.nx / span.x, grid.ny / span.y, grid.nz / span.z};

    const int64_t ncells = (int64_t)grid.nx * grid.ny * grid.nz;
    std::vector<int32_t> counts(ncells, 0);
    auto cell_range = [&](int64_t t, int lo[3], int hi[3]) {
      Vec3 tlo{1e300, 1e300, 1e300}, thi{-1e300, -1e300, -1e300};
      for (int k = 0; k < 4; ++k) {
        const Vec3 p = vert(tet2vert[t * 4 + k]);
        tlo.x = std::min(tlo.x, p.x); thi.x = std::max(thi.x, p.x);
        tlo.y = std::min(tlo.y, p.y); thi.y = std::max(thi.y, p.y);
        tlo.z = std::min(tlo.z, p.z); thi.z = std::max(thi.z, p.z);
      }
      lo[0] = std::clamp((int)((tlo.x - grid.lo.x) * grid.inv_h.x), 0, grid.nx - 1);
      lo[1] = std::clamp((int)((tlo.y - grid.lo.y) * grid.inv_h.y), 0, grid.ny - 1);
      lo[2] = std::clamp((int)((tlo.z - grid.lo.z) * grid.inv_h.z), 0, grid.nz - 1);
      hi[0] = std::clamp((int)((thi.x - grid.lo.x) * grid.inv_h.x), 0, grid.nx - 1);
      hi[1] = std::clamp((int)((thi.y - grid.lo.y) * grid.inv_h.y), 0, grid.ny - 1);
      hi[2] = std::clamp((int)((thi.z - grid.lo.z) * grid.inv_h.z), 0, grid.nz - 1);
    };
    for (int64_t t = 0; t < nelems; ++t) {
      int lo[3], hi[3];
      cell_range(t, lo, hi);
      for (int cz = lo[2]; cz <= hi[2]; ++cz)
        for (int cy = lo[1]; cy <= hi[1]; ++cy)
          for (int cx = lo[0]; cx <= hi[0]; ++cx)
            counts[((int64_t)cz * grid.ny + cy) * grid.nx + cx]++;
    }
    grid.cell_start.assign(ncells + 1, 0);
    for (int64_t c = 0; c < ncells; ++c)
      grid.cell_start[c + 1] = grid.cell_start[c] + counts[c];
    grid.cell_tets.resize(grid.cell_start[ncells]);
    std::vector<int32_t> cursor(grid.cell_start.begin(), grid.cell_start.end() - 1);
    for (int64_t t = 0; t < nelems; ++t) {
      int lo[3], hi[3];
      cell_range(t, lo, hi);
      for (int cz = lo[2]; cz <= hi[2]; ++cz)
        for (int cy = lo[1]; cy <= hi[1]; ++cy)
          for (int cx = lo[0]; cx <= hi[0]; ++cx) {
            const int64_t c = ((int64_t)cz * grid.ny + cy) * grid.nx + cx;
            grid.cell_tets[cursor[c]++] = (int32_t)t;
          }
    }
  }
}

int32_t Mesh::locate(Vec3 p, double tol, bool *used_loose) const {
  const int cx = std::clamp((int)((p.x - grid.lo.x) * grid.inv_h.x), 0, grid.nx - 1);
  const int cy = std::clamp((int)((p.y - grid.lo.y) * grid.inv_h.y), 0, grid.ny - 1);
  const int cz = std::clamp((int)((p.z - grid.lo.z) * grid.inv_h.z), 0, grid.nz - 1);
  const int64_t c = ((int64_t)cz * grid.ny + cy) * grid.nx + cx;
  for (int32_t i = grid.cell_start[c]; i < grid.cell_start[c + 1]; ++i)
    if (contains(grid.cell_tets[i], p, tol)) return grid.cell_tets[i];
  // Retry with a looser tolerance before declaring the point outside: points
  // exactly on cell-boundary faces can fail the strict test in every listed
  // tet by a few ulps.
  for (int32_t i = grid.cell_start[c]; i < grid.cell_start[c + 1]; ++i)
    if (contains(grid.cell_tets[i], p, tol * 1e4)) {
      if (used_loose) *used_loose = true;
      return grid.cell_tets[i];
    }
  return -1;
}

Mesh build_box(int nx, int ny, int nz, double lx, double ly, double lz) {
  Mesh m;
  const int vx = nx + 1, vy = ny + 1, vz = nz + 1;
  m.nverts = (int64_t)vx * vy * vz;
  m.coords.resize(m.nverts * 3);
  for (int k = 0; k < vz; ++k)
    for (int j = 0; j < vy; ++j)
      for (int i = 0; i < vx; ++i) {
        const int64_t v = ((int64_t)k * vy + j) * vx + i;
        m.coords[v * 3] = lx * i / nx;
        m.coords[v * 3 + 1] = ly * j / ny;
        m.coords[v * 3 + 2] = lz * k / nz;
      }
  auto vid = [&](int i, int j, int k) -> int32_t {
    return (int32_t)(((int64_t)k * vy + j) * vx + i);
  };
  // 6-tet cut of each cell around the main diagonal c0->c7.  Element ordering
  // within a cell matches Omega_h::build_box for the unit cube (derived from
  // the reference tests): element k covers the region where the local
  // coordinates are ordered as below (u=x-x0 etc. scaled to the cell):
  //   el0: y>=x>=z  {0,2,3,7}   el1: y>=z>=x  {0,2,6,7}
  //   el2: z>=y>=x  {0,4,6,7}   el3: z>=x>=y  {0,4,5,7}
  //   el4: x>=z>=y  {0,1,5,7}   el5: x>=y>=z  {0,1,3,7}
  // Corner numbering: bit0=x, bit1=y, bit2=z.
  static const int chain[6][2] = {{2, 3}, {2, 6}, {4, 6}, {4, 5}, {1, 5}, {1, 3}};
  m.nelems = (int64_t)nx * ny * nz * 6;
  m.tet2vert.resize(m.nelems * 4);
  int64_t t = 0;
  for (int k = 0; k < nz; ++k)
    for (int j = 0; j < ny; ++j)
      for (int i = 0; i < nx; ++i) {
        int32_t corner[8];
        for (int b = 0; b < 8; ++b)
          corner[b] = vid(i + (b & 1), j + ((b >> 1) & 1), k + ((b >> 2) & 1));
        for (int e = 0; e < 6; ++e, ++t) {
          int32_t *tv = &m.tet2vert[t * 4];
          tv[0] = corner[0];
          tv[1] = corner[chain[e][0]];
          tv[2] = corner[chain[e][1]];
          tv[3] = corner[7];
        }
      }
  m.finalize(); // fixes orientation where the chain winds negatively
  return m;
}

Mesh mesh_from_arrays(int64_t nverts, const double *coords, int64_t nelems,
                      const int32_t *tets) {
  Mesh m;
  m.nverts = nverts;
  m.nelems = nelems;
  m.coords.assign(coords, coords + nverts * 3);
  m.tet2vert.assign(tets, tets + nelems * 4);
  m.finalize();
  return m;
}

} // namespace pumitally
