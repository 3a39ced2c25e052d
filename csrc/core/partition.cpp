// Element-ownership domain decomposition (see mesh.h for the contract).
#include "mesh.h"

#include <algorithm>
#include <cstdint>
#include <numeric>
#include <stdexcept>

namespace pumitally {

namespace {
uint64_t morton3(uint32_t x, uint32_t y, uint32_t z) {
  auto spread = [](uint64_t v) {
    v &= 0x1fffff;
    v = (v | v << 32) & 0x1f00000000ffffull;
    v = (v | v << 16) & 0x1f0000ff0000ffull;
    v = (v | v << 8) & 0x100f00f00f00f00full;
    v = (v | v << 4) & 0x10c30c30c30c30c3ull;
    v = (v | v << 2) & 0x1249249249249249ull;
    return v;
  };
  return spread(x) | (spread(y) << 1) | (spread(z) << 2);
}
} // namespace

std::vector<int32_t> partition_morton(const Mesh &m, int nparts,
                                      const double *weights) {
  if (nparts < 1) throw std::runtime_error("nparts must be >= 1");
  if (m.nelems < nparts)
    throw std::runtime_error(
        "partition_morton: mesh has " + std::to_string(m.nelems) +
        " elements but " + std::to_string(nparts) +
        " parts were requested; every part must own at least one element");
  std::vector<int64_t> order(m.nelems);
  std::iota(order.begin(), order.end(), 0);
  if (nparts > 1) {
    const Vec3 ext = m.bbox_hi - m.bbox_lo;
    const double sx = ext.x > 0 ? 2097151.0 / ext.x : 0.0;
    const double sy = ext.y > 0 ? 2097151.0 / ext.y : 0.0;
    const double sz = ext.z > 0 ? 2097151.0 / ext.z : 0.0;
    std::vector<uint64_t> key(m.nelems);
    for (int64_t t = 0; t < m.nelems; ++t) {
      const Vec3 c = m.centroid((int32_t)t);
      key[t] = morton3((uint32_t)((c.x - m.bbox_lo.x) * sx),
                       (uint32_t)((c.y - m.bbox_lo.y) * sy),
                       (uint32_t)((c.z - m.bbox_lo.z) * sz));
    }
    std::sort(order.begin(), order.end(),
              [&](int64_t a, int64_t b) { return key[a] < key[b]; });
  }
  std::vector<int32_t> owners(m.nelems);
  if (!weights) {
    for (int64_t i = 0; i < m.nelems; ++i)
      owners[order[i]] = (int32_t)((i * nparts) / m.nelems);
    return owners;
  }
  // Work-weighted split: equal prefix-sum chunks along the Morton curve.
  // Weights are per-element work estimates (e.g. the previous batch's raw
  // flux + epsilon); zero/negative entries get a small floor so every
  // element stays assignable and parts stay contiguous on the curve.
  double total = 0.0;
  double wmax = 0.0;
  for (int64_t t = 0; t < m.nelems; ++t)
    wmax = std::max(wmax, weights[t] > 0 ? weights[t] : 0.0);
  const double floor_w = wmax > 0 ? wmax * 1e-6 : 1.0;
  std::vector<double> w(m.nelems);
  for (int64_t t = 0; t < m.nelems; ++t) {
    w[t] = weights[t] > floor_w ? weights[t] : floor_w;
    total += w[t];
  }
  // Split the curve where the prefix weight-midpoint crosses k*per: element
  // i belongs to part k iff k*per <= mid_i < (k+1)*per (mid_i increasing).
  // The clamp passes then guarantee every part owns at least one element
  // even under extreme skew (one element holding > total/nparts weight),
  // so no rank ever receives an empty submesh.
  const double per = total / nparts;
  std::vector<int64_t> split((size_t)nparts + 1);
  split[0] = 0;
  split[nparts] = m.nelems;
  {
    double acc = 0.0;
    int64_t i = 0;
    for (int k = 1; k < nparts; ++k) {
      const double target = k * per;
      while (i < m.nelems && acc + 0.5 * w[order[i]] < target) {
        acc += w[order[i]];
        ++i;
      }
      split[k] = i;
    }
  }
  for (int k = 1; k < nparts; ++k)
    split[k] = std::max(split[k], (int64_t)k);
  for (int k = nparts - 1; k >= 1; --k)
    split[k] = std::min(split[k], split[k + 1] - 1);
  for (int k = 0; k < nparts; ++k)
    for (int64_t j = split[k]; j < split[k + 1]; ++j)
      owners[order[j]] = (int32_t)k;
  return owners;
}

SubMesh extract_submesh(const Mesh &m, const std::vector<int32_t> &owners,
                        int part, int ghost_rings) {
  if ((int64_t)owners.size() != m.nelems)
    throw std::runtime_error("owners size mismatch");
  SubMesh sub;
  std::vector<char> in_sub(m.nelems, 0);
  for (int64_t g = 0; g < m.nelems; ++g)
    if (owners[g] == part) {
      sub.elem_l2g.push_back(g);
      in_sub[g] = 1;
    }
  // grow ghost rings by face adjacency
  std::vector<int64_t> frontier(sub.elem_l2g);
  for (int r = 0; r < ghost_rings; ++r) {
    std::vector<int64_t> next;
    for (int64_t g : frontier)
      for (int f = 0; f < 4; ++f) {
        const int32_t nb = m.nbr[g * 4 + f];
        if (nb >= 0 && !in_sub[nb]) {
          in_sub[nb] = 1;
          sub.elem_l2g.push_back(nb);
          next.push_back(nb);
        }
      }
    frontier.swap(next);
  }
  std::sort(sub.elem_l2g.begin(), sub.elem_l2g.end());
  const int64_t ne = (int64_t)sub.elem_l2g.size();

  // local vertex numbering, ascending in global id (keeps canonical face
  // planes bitwise-identical to the full mesh)
  std::vector<int32_t> used;
  used.reserve(ne * 4);
  for (int64_t t = 0; t < ne; ++t)
    for (int k = 0; k < 4; ++k) used.push_back(m.tet2vert[sub.elem_l2g[t] * 4 + k]);
  std::sort(used.begin(), used.end());
  used.erase(std::unique(used.begin(), used.end()), used.end());
  std::vector<int32_t> g2l_vert(m.nverts, -1);
  for (size_t i = 0; i < used.size(); ++i) g2l_vert[used[i]] = (int32_t)i;

  Mesh &local = sub.local;
  local.nverts = (int64_t)used.size();
  local.coords.resize(local.nverts * 3);
  for (size_t i = 0; i < used.size(); ++i)
    for (int k = 0; k < 3; ++k)
      local.coords[i * 3 + k] = m.coords[(int64_t)used[i] * 3 + k];
  local.nelems = ne;
  local.tet2vert.resize(ne * 4);
  for (int64_t t = 0; t < ne; ++t)
    for (int k = 0; k < 4; ++k)
      local.tet2vert[t * 4 + k] = g2l_vert[m.tet2vert[sub.elem_l2g[t] * 4 + k]];
  local.finalize(); // positive orientation preserved -> no vertex swaps

  // Mark cross-part faces.  After finalize(), any local face with nbr==-1
  // is either a true mesh boundary or a partition cut; consult the global
  // adjacency to tell them apart.
  std::vector<int64_t> g2l_elem(m.nelems, -1);
  for (int64_t t = 0; t < ne; ++t) g2l_elem[sub.elem_l2g[t]] = t;
  for (int64_t t = 0; t < ne; ++t) {
    const int64_t g = sub.elem_l2g[t];
    for (int f = 0; f < 4; ++f) {
      if (local.nbr[t * 4 + f] != -1) continue;
      const int32_t gn = m.nbr[g * 4 + f];
      if (gn == -1) {
        // true boundary: reflective bit, periodic pair, or vacuum
        const int64_t gface = g * 4 + f;
        if (m.face_is_reflective(gface))
          local.set_face_reflective(t * 4 + f);
        if (!m.periodic_idx.empty() && m.periodic_idx[gface] >= 0) {
          const int32_t kg = m.periodic_idx[gface];
          const int32_t ge = m.periodic_elem[kg];
          const double *sh = m.periodic_shift.data() + (int64_t)kg * 3;
          if (g2l_elem[ge] >= 0) {
            // partner element is in this submesh (owned or ghost):
            // local periodic restart, exactly as on the full mesh
            if (local.periodic_idx.empty())
              local.periodic_idx.assign(ne * 4, -1);
            const int32_t kl = (int32_t)local.periodic_elem.size();
            local.periodic_elem.push_back((int32_t)g2l_elem[ge]);
            local.periodic_shift.insert(local.periodic_shift.end(),
                                        {sh[0], sh[1], sh[2]});
            local.periodic_idx[t * 4 + f] = kl;
          } else {
            // partner lives on another part: hand off with the
            // translation (the driver applies it to pos AND dest)
            const int32_t k = (int32_t)sub.foreign_gid.size();
            sub.foreign_gid.push_back(ge);
            sub.foreign_owner.push_back(owners[ge]);
            sub.foreign_shift.insert(sub.foreign_shift.end(),
                                     {sh[0], sh[1], sh[2]});
            local.nbr[t * 4 + f] = -(2 + k);
          }
        }
        continue;
      }
      const int32_t k = (int32_t)sub.foreign_gid.size();
      sub.foreign_gid.push_back(gn);
      sub.foreign_owner.push_back(owners[gn]);
      sub.foreign_shift.insert(sub.foreign_shift.end(), {0.0, 0.0, 0.0});
      local.nbr[t * 4 + f] = -(2 + k);
    }
  }
  return sub;
}

} // namespace pumitally
