// Python bindings: pumiumtally_amd._core
//
// Thin numpy-in/numpy-out layer over the C++ core.  No torch dependency --
// the distributed driver (pumiumtally_amd.parallel) uses torch.distributed
// on top of these bindings for RCCL collectives.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "../api/PumiTally.h"
#include "../comm/comm.h"
#include "../core/engine.h"
#include "../core/partition_engine.h"

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstring>
#include <memory>
#include <thread>
#include <vector>

namespace py = pybind11;
using namespace pumitally;

namespace {

bool have_gpu() {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) {
    (void)hipGetLastError();
    return false;
  }
  return n > 0;
}

py::array_t<double> vec_to_np(std::vector<double> v) {
  auto out = py::array_t<double>(v.size());
  std::memcpy(out.mutable_data(), v.data(), v.size() * sizeof(double));
  return out;
}

struct PyEngine {
  std::unique_ptr<Engine> eng;
  bool gpu = false;

  PyEngine(const Mesh &mesh, int64_t n, const std::string &device,
           int ngroups = 1, int nscores = 1) {
    if (device == "cpu") {
      eng = make_cpu_engine(mesh, n, ngroups, nscores);
    } else {
      int ordinal = 0;
      if (device.rfind("cuda:", 0) == 0) ordinal = std::stoi(device.substr(5));
      else if (device.rfind("gpu:", 0) == 0) ordinal = std::stoi(device.substr(4));
      else if (device != "cuda" && device != "gpu" && device != "auto")
        throw std::runtime_error("device must be cpu/cuda[:N]/auto");
      eng = make_gpu_engine(mesh, n, ordinal, ngroups, nscores);
      if (eng) {
        gpu = true;
      } else if (device == "auto") {
        eng = make_cpu_engine(mesh, n, ngroups, nscores);
      } else {
        // Fail loudly: a GPU was requested but none is usable.  GPU tests
        // must never fall back silently to the CPU oracle.
        throw std::runtime_error("no usable HIP device for device=" + device);
      }
    }
  }
};

// Pinned host allocation exposed as a numpy array (zero-copy H2D staging for
// the bench / host transport codes).  Falls back to pageable memory when no
// GPU is present so the same code runs on CPU-only machines.
py::array pinned_array(py::object shape_obj, const std::string &dtype) {
  std::vector<py::ssize_t> shape;
  if (py::isinstance<py::int_>(shape_obj)) {
    shape.push_back(shape_obj.cast<py::ssize_t>());
  } else {
    for (auto s : shape_obj.cast<py::sequence>())
      shape.push_back(s.cast<py::ssize_t>());
  }
  py::dtype dt = py::dtype(dtype);
  py::ssize_t count = 1;
  for (auto s : shape) count *= s;
  const size_t bytes = (size_t)count * dt.itemsize();
  void *p = nullptr;
  if (have_gpu()) {
    if (hipHostMalloc(&p, bytes ? bytes : 1, hipHostMallocDefault) != hipSuccess)
      throw std::runtime_error("hipHostMalloc failed");
    py::capsule owner(p, [](void *q) { (void)hipHostFree(q); });
    return py::array(dt, shape, p, owner);
  }
  p = ::malloc(bytes ? bytes : 1);
  py::capsule owner(p, [](void *q) { ::free(q); });
  return py::array(dt, shape, p, owner);
}

} // namespace

PYBIND11_MODULE(_core, m) {
  m.doc() = "MI355X-native unstructured-mesh track-length tally engine";

  py::class_<Mesh>(m, "Mesh")
      .def_property_readonly("nelems", [](const Mesh &m_) { return m_.nelems; })
      .def_property_readonly("nverts", [](const Mesh &m_) { return m_.nverts; })
      .def_property_readonly("coords",
                             [](const Mesh &m_) {
                               auto a = py::array_t<double>({m_.nverts, (int64_t)3});
                               std::memcpy(a.mutable_data(), m_.coords.data(),
                                           m_.coords.size() * sizeof(double));
                               return a;
                             })
      .def_property_readonly("tet2vert",
                             [](const Mesh &m_) {
                               auto a = py::array_t<int32_t>({m_.nelems, (int64_t)4});
                               std::memcpy(a.mutable_data(), m_.tet2vert.data(),
                                           m_.tet2vert.size() * sizeof(int32_t));
                               return a;
                             })
      .def_property_readonly("neighbors",
                             [](const Mesh &m_) {
                               auto a = py::array_t<int32_t>({m_.nelems, (int64_t)4});
                               std::memcpy(a.mutable_data(), m_.nbr.data(),
                                           m_.nbr.size() * sizeof(int32_t));
                               return a;
                             })
      .def_property_readonly("volumes",
                             [](const Mesh &m_) { return vec_to_np(m_.volumes); })
      .def("centroid",
           [](const Mesh &m_, int32_t t) {
             const Vec3 c = m_.centroid(t);
             return py::make_tuple(c.x, c.y, c.z);
           })
      .def("locate",
           [](const Mesh &m_, py::array_t<double, py::array::c_style | py::array::forcecast> pts) {
             const int64_t n = pts.size() / 3;
             auto out = py::array_t<int32_t>(n);
             const double *p = pts.data();
             int32_t *o = out.mutable_data();
             const double tol = 1e-10 * norm(m_.bbox_hi - m_.bbox_lo);
             py::gil_scoped_release nogil;
             auto range = [&](int64_t lo, int64_t hi) {
               for (int64_t i = lo; i < hi; ++i)
                 o[i] = m_.locate({p[i * 3], p[i * 3 + 1], p[i * 3 + 2]}, tol);
             };
             const unsigned hw = std::thread::hardware_concurrency();
             if (n >= 16384 && hw > 1) {
               const int nthreads = (int)std::min<unsigned>(hw, 64);
               const int64_t per = (n + nthreads - 1) / nthreads;
               std::vector<std::thread> workers;
               for (int t = 0; t < nthreads; ++t)
                 workers.emplace_back([&, t] {
                   range(t * per, std::min<int64_t>(n, (t + 1) * per));
                 });
               for (auto &w : workers) w.join();
             } else {
               range(0, n);
             }
             return out;
           })
      .def("write_vtk",
           [](const Mesh &m_, const std::string &path) { write_vtk(path, m_, {}); })
      .def("write_vtk_fields",
           [](const Mesh &m_, const std::string &path, py::list fields) {
             std::vector<std::pair<std::string, std::vector<double>>> cd;
             for (auto item : fields) {
               auto t = item.cast<py::tuple>();
               auto name = t[0].cast<std::string>();
               auto arr = t[1].cast<py::array_t<double, py::array::c_style | py::array::forcecast>>();
               cd.emplace_back(name, std::vector<double>(arr.data(), arr.data() + arr.size()));
             }
             write_vtk(path, m_, cd);
           })
      .def("write_osh",
           [](const Mesh &m_, const std::string &dir) { write_osh(dir, m_); })
      .def("boundary_faces",
           // (face_indices=elem*4+f, centroids (k,3), outward normals (k,3))
           // of every boundary face, for marking per-face BCs.
           [](const Mesh &m_) {
             std::vector<int64_t> ids;
             for (int64_t t = 0; t < m_.nelems; ++t)
               for (int f = 0; f < 4; ++f)
                 if (m_.nbr[t * 4 + f] == -1) ids.push_back(t * 4 + f);
             const int64_t k = (int64_t)ids.size();
             auto fid = py::array_t<int64_t>(k);
             auto cen = py::array_t<double>({k, (int64_t)3});
             auto nor = py::array_t<double>({k, (int64_t)3});
             for (int64_t i = 0; i < k; ++i) {
               const int64_t t = ids[i] / 4;
               const int f = (int)(ids[i] % 4);
               fid.mutable_data()[i] = ids[i];
               Vec3 c{0, 0, 0};
               for (int j = 0; j < 3; ++j)
                 c = c + m_.vert(m_.tet2vert[t * 4 + kFaceVerts[f][j]]);
               c = (1.0 / 3.0) * c;
               const Plane &pl = m_.planes[ids[i]]; // inward-positive
               cen.mutable_data()[i * 3] = c.x;
               cen.mutable_data()[i * 3 + 1] = c.y;
               cen.mutable_data()[i * 3 + 2] = c.z;
               nor.mutable_data()[i * 3] = -pl.nx;
               nor.mutable_data()[i * 3 + 1] = -pl.ny;
               nor.mutable_data()[i * 3 + 2] = -pl.nz;
             }
             return py::make_tuple(fid, cen, nor);
           })
      .def("face_is_reflective",
           [](const Mesh &m_, int64_t fid) { return m_.face_is_reflective(fid); })
      .def("set_reflective_faces",
           [](Mesh &m_, py::array_t<int64_t, py::array::c_style | py::array::forcecast> fids) {
             for (py::ssize_t i = 0; i < fids.size(); ++i)
               m_.set_face_reflective(fids.data()[i]);
           })
      .def("set_periodic_faces",
           // Pair every boundary face in A with the boundary face in B at
           // centroid_A + translation; walks leaving through one re-enter
           // through the other (remaining segment translated).
           [](Mesh &m_, py::array_t<int64_t, py::array::c_style | py::array::forcecast> a,
              py::array_t<int64_t, py::array::c_style | py::array::forcecast> b,
              py::array_t<double, py::array::c_style | py::array::forcecast> t,
              double tol) {
             if (t.size() != 3)
               throw std::runtime_error("translation must have 3 components");
             m_.set_periodic_faces(
                 std::vector<int64_t>(a.data(), a.data() + a.size()),
                 std::vector<int64_t>(b.data(), b.data() + b.size()),
                 Vec3{t.data()[0], t.data()[1], t.data()[2]}, tol);
           },
           py::arg("faces_a"), py::arg("faces_b"), py::arg("translation"),
           py::arg("tol") = -1.0)
      .def_property_readonly("has_periodic",
                             [](const Mesh &m_) { return m_.has_periodic(); });

  py::class_<SubMesh>(m, "SubMesh")
      .def_property_readonly("local", [](const SubMesh &s) -> const Mesh & { return s.local; },
                             py::return_value_policy::reference_internal)
      .def_property_readonly("elem_l2g",
                             [](const SubMesh &s) {
                               auto a = py::array_t<int64_t>(s.elem_l2g.size());
                               std::memcpy(a.mutable_data(), s.elem_l2g.data(),
                                           s.elem_l2g.size() * 8);
                               return a;
                             })
      .def_property_readonly("foreign_gid",
                             [](const SubMesh &s) {
                               auto a = py::array_t<int64_t>(s.foreign_gid.size());
                               std::memcpy(a.mutable_data(), s.foreign_gid.data(),
                                           s.foreign_gid.size() * 8);
                               return a;
                             })
      .def_property_readonly("foreign_shift",
                             [](const SubMesh &s) {
                               auto a = py::array_t<double>(
                                   {(py::ssize_t)(s.foreign_shift.size() / 3),
                                    (py::ssize_t)3});
                               std::memcpy(a.mutable_data(),
                                           s.foreign_shift.data(),
                                           s.foreign_shift.size() * 8);
                               return a;
                             })
      .def_property_readonly("foreign_owner", [](const SubMesh &s) {
        auto a = py::array_t<int32_t>(s.foreign_owner.size());
        std::memcpy(a.mutable_data(), s.foreign_owner.data(),
                    s.foreign_owner.size() * 4);
        return a;
      });

  m.def("partition_morton",
        [](const Mesh &m_, int nparts, py::object weights_obj) {
          const double *wp = nullptr;
          py::array_t<double, py::array::c_style | py::array::forcecast> w;
          if (!weights_obj.is_none()) {
            w = weights_obj.cast<
                py::array_t<double, py::array::c_style | py::array::forcecast>>();
            if ((int64_t)w.size() != m_.nelems)
              throw std::runtime_error("partition_morton: weights size != nelems");
            wp = w.data();
          }
          auto owners = partition_morton(m_, nparts, wp);
          auto a = py::array_t<int32_t>(owners.size());
          std::memcpy(a.mutable_data(), owners.data(), owners.size() * 4);
          return a;
        },
        py::arg("mesh"), py::arg("nparts"), py::arg("weights") = py::none());
  m.def("extract_submesh",
        [](const Mesh &m_, py::array_t<int32_t, py::array::c_style | py::array::forcecast> owners,
           int part, int ghost_rings) {
          std::vector<int32_t> o(owners.data(), owners.data() + owners.size());
          return extract_submesh(m_, o, part, ghost_rings);
        },
        py::arg("mesh"), py::arg("owners"), py::arg("part"),
        py::arg("ghost_rings") = 0);

  m.def("build_box", &build_box, py::arg("nx"), py::arg("ny"), py::arg("nz"),
        py::arg("lx") = 1.0, py::arg("ly") = 1.0, py::arg("lz") = 1.0);
  m.def("read_mesh", &read_mesh);
  m.def("read_gmsh", &read_gmsh);
  m.def("read_osh", &read_osh);
  m.def("mesh_from_arrays",
        [](py::array_t<double, py::array::c_style | py::array::forcecast> coords,
           py::array_t<int32_t, py::array::c_style | py::array::forcecast> tets) {
          return mesh_from_arrays(coords.size() / 3, coords.data(),
                                  tets.size() / 4, tets.data());
        });
  m.def("have_gpu", &have_gpu);

  // Stateful domain-decomposed engine (csrc/hip/partition_engine.hip):
  // particles stay resident on their owner rank between steps; the comm
  // (when world > 1) is the library's own TCP/RCCL layer, created from
  // the torchrun-compatible env.
  struct PyPartEngine {
    std::unique_ptr<Comm> comm;
    std::unique_ptr<PartitionedEngine> pe;
  };
  py::class_<PyPartEngine>(m, "PartitionedEngine")
      .def(py::init([](const Mesh &mesh, int64_t n_global,
                       const std::string &device, int ngroups, int nscores,
                       py::object owners, int ghost_rings) {
             auto self = std::make_unique<PyPartEngine>();
             const EnvComm env = comm_env();
             self->comm =
                 make_comm_from_env(device != "cpu", env.local_rank);
             const int rank = self->comm ? self->comm->rank() : 0;
             const int world = self->comm ? self->comm->world() : 1;
             const int32_t *op = nullptr;
             py::array_t<int32_t, py::array::c_style | py::array::forcecast>
                 oarr;
             if (!owners.is_none()) {
               oarr = owners.cast<py::array_t<
                   int32_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)oarr.size() != mesh.nelems)
                 throw std::runtime_error("owners size != nelems");
               op = oarr.data();
             }
             std::string dev = device;
             if (dev == "auto" || dev.empty())
               dev = have_gpu() ? ("cuda:" + std::to_string(env.local_rank))
                                : "cpu";
             self->pe = make_partitioned_engine(mesh, n_global,
                                                self->comm.get(), rank,
                                                world, dev, ngroups, nscores,
                                                op, ghost_rings);
             return self;
           }),
           py::arg("mesh"), py::arg("n_global"), py::arg("device") = "auto",
           py::arg("ngroups") = 1, py::arg("nscores") = 1,
           py::arg("owners") = py::none(), py::arg("ghost_rings") = 1)
      .def_property_readonly("rank",
                             [](const PyPartEngine &s) { return s.pe->rank(); })
      .def_property_readonly(
          "world", [](const PyPartEngine &s) { return s.pe->world(); })
      .def_property_readonly(
          "num_particles",
          [](const PyPartEngine &s) { return s.pe->num_particles(); })
      .def_property_readonly(
          "resident", [](const PyPartEngine &s) { return s.pe->resident(); })
      .def("localize",
           [](PyPartEngine &s,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  origins) {
             if ((int64_t)origins.size() != s.pe->num_particles() * 3)
               throw std::runtime_error("localize: size must be 3*n_global");
             py::gil_scoped_release ng;
             s.pe->localize(origins.data(), s.pe->num_particles());
           })
      .def("step",
           [](PyPartEngine &s,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  dest,
              py::array_t<int8_t, py::array::c_style | py::array::forcecast>
                  flying,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  weights,
              py::object origin, py::object groups, py::object responses) {
             const int64_t n = s.pe->num_particles();
             if ((int64_t)dest.size() != n * 3 ||
                 (int64_t)flying.size() != n ||
                 (int64_t)weights.size() != n)
               throw std::runtime_error("step: array size mismatch");
             const double *op = nullptr;
             py::array_t<double, py::array::c_style | py::array::forcecast>
                 oarr;
             if (!origin.is_none()) {
               oarr = origin.cast<py::array_t<
                   double, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)oarr.size() != n * 3)
                 throw std::runtime_error("step: origin size mismatch");
               op = oarr.data();
             }
             const uint16_t *gp = nullptr;
             py::array_t<uint16_t, py::array::c_style | py::array::forcecast>
                 garr;
             if (!groups.is_none()) {
               garr = groups.cast<py::array_t<
                   uint16_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)garr.size() != n)
                 throw std::runtime_error("step: groups size mismatch");
               gp = garr.data();
             }
             const double *rp = nullptr;
             py::array_t<double, py::array::c_style | py::array::forcecast>
                 rarr;
             if (!responses.is_none()) {
               rarr = responses.cast<py::array_t<
                   double, py::array::c_style | py::array::forcecast>>();
               rp = rarr.data();
             }
             py::gil_scoped_release ng;
             s.pe->step(dest.data(), flying.data(), weights.data(), n, op,
                        gp, rp);
           },
           py::arg("dest"), py::arg("flying"), py::arg("weights"),
           py::arg("origin") = py::none(), py::arg("groups") = py::none(),
           py::arg("responses") = py::none())
      .def("flux_global",
           [](PyPartEngine &s) {
             std::vector<double> f;
             {
               py::gil_scoped_release ng;
               f = s.pe->flux_global();
             }
             return vec_to_np(std::move(f));
           })
      .def("stats",
           [](const PyPartEngine &s) {
             const EngineStats &st = s.pe->stats();
             py::dict d;
             d["lost_particles"] = st.lost_particles;
             d["moves"] = st.moves;
             d["relocated"] = st.relocated;
             d["loose_localizations"] = st.loose_localizations;
             return d;
           })
      .def("resident_mask",
           [](const PyPartEngine &s) {
             auto v = s.pe->resident_mask();
             py::array_t<uint8_t> out(v.size());
             std::memcpy(out.mutable_data(), v.data(), v.size());
             return out;
           })
      .def("positions",
           [](const PyPartEngine &s) { return vec_to_np(s.pe->positions()); })
      .def("elem_ids",
           [](const PyPartEngine &s) {
             auto v = s.pe->elem_ids();
             py::array_t<int32_t> out(v.size());
             std::memcpy(out.mutable_data(), v.data(), v.size() * 4);
             return out;
           })
      // decomposition-independent state transfer (checkpoint/resume and
      // repartitioning -- see partition_engine.h)
      .def("elem_ids_global",
           [](const PyPartEngine &s) {
             auto v = s.pe->elem_ids_global();
             py::array_t<int32_t> out(v.size());
             std::memcpy(out.mutable_data(), v.data(), v.size() * 4);
             return out;
           })
      .def("escaped_mask",
           [](const PyPartEngine &s) {
             auto v = s.pe->escaped_mask();
             py::array_t<uint8_t> out(v.size());
             std::memcpy(out.mutable_data(), v.data(), v.size());
             return out;
           })
      .def("set_state",
           [](PyPartEngine &s,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  pos,
              py::array_t<int32_t, py::array::c_style | py::array::forcecast>
                  gelem,
              py::array_t<uint8_t, py::array::c_style | py::array::forcecast>
                  escaped) {
             const int64_t n = s.pe->num_particles();
             if ((int64_t)pos.size() != n * 3 ||
                 (int64_t)gelem.size() != n ||
                 (int64_t)escaped.size() != n)
               throw std::runtime_error("set_state: size mismatch");
             py::gil_scoped_release ng;
             s.pe->set_state(pos.data(), gelem.data(), escaped.data(), n);
           })
      .def("synchronize", [](PyPartEngine &s) { s.pe->synchronize(); })
      // coupled-host path: resident_list() snapshots this rank's
      // particles; step_local consumes arrays in exactly that order
      .def("resident_list",
           [](PyPartEngine &s) {
             std::vector<int64_t> v;
             {
               py::gil_scoped_release ng;
               v = s.pe->resident_list();
             }
             auto out = py::array_t<int64_t>(v.size());
             std::memcpy(out.mutable_data(), v.data(), v.size() * 8);
             return out;
           })
      .def("step_local",
           [](PyPartEngine &s,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  dest,
              py::array_t<int8_t, py::array::c_style | py::array::forcecast>
                  flying,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  weights,
              py::object origin, py::object groups, py::object responses) {
             const int64_t nl = (int64_t)flying.size();
             if ((int64_t)dest.size() != nl * 3 ||
                 (int64_t)weights.size() != nl)
               throw std::runtime_error("step_local: array size mismatch");
             const double *op = nullptr;
             py::array_t<double, py::array::c_style | py::array::forcecast>
                 oarr;
             if (!origin.is_none()) {
               oarr = origin.cast<py::array_t<
                   double, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)oarr.size() != nl * 3)
                 throw std::runtime_error("step_local: origin size mismatch");
               op = oarr.data();
             }
             const uint16_t *gp = nullptr;
             py::array_t<uint16_t, py::array::c_style | py::array::forcecast>
                 garr;
             if (!groups.is_none()) {
               garr = groups.cast<py::array_t<
                   uint16_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)garr.size() != nl)
                 throw std::runtime_error("step_local: groups size mismatch");
               gp = garr.data();
             }
             const double *rp = nullptr;
             py::array_t<double, py::array::c_style | py::array::forcecast>
                 rarr;
             if (!responses.is_none()) {
               rarr = responses.cast<py::array_t<
                   double, py::array::c_style | py::array::forcecast>>();
               rp = rarr.data();
             }
             py::gil_scoped_release ng;
             s.pe->step_local(dest.data(), flying.data(), weights.data(), nl,
                              op, gp, rp);
           },
           py::arg("dest"), py::arg("flying"), py::arg("weights"),
           py::arg("origin") = py::none(), py::arg("groups") = py::none(),
           py::arg("responses") = py::none())
      // comm passthroughs so a driver (bench.py --partitioned) needs no
      // second communication stack
      .def("barrier",
           [](PyPartEngine &s) {
             if (s.comm) {
               py::gil_scoped_release ng;
               s.comm->barrier();
             }
           })
      .def_property_readonly(
          "comm_kind",
          [](PyPartEngine &s) -> std::string {
            if (!s.comm) return "local";
            return s.comm->has_device_collectives() ? "rccl" : "tcp";
          })
      .def("allreduce_max",
           [](PyPartEngine &s,
              py::array_t<double, py::array::c_style | py::array::forcecast>
                  a) {
             if (s.comm) {
               py::gil_scoped_release ng;
               s.comm->allreduce_max(a.mutable_data(), (int64_t)a.size());
             }
             return a; // forcecast may have copied; reduced values are here
           });

  // The library's own communication layer (csrc/comm): RCCL over xGMI on
  // GPU, TCP fallback on CPU -- no torch, no MPI.  Python surface for
  // bench.py --native-comm and the torch-free distributed path.
  py::class_<Comm>(m, "NativeComm")
      .def_property_readonly("rank", &Comm::rank)
      .def_property_readonly("world", &Comm::world)
      .def("barrier", [](Comm &c) { py::gil_scoped_release ng; c.barrier(); })
      .def("allreduce_sum",
           [](Comm &c, py::array_t<double, py::array::c_style |
                                               py::array::forcecast> a) {
             py::gil_scoped_release ng;
             c.allreduce_sum(a.mutable_data(), (int64_t)a.size());
           })
      .def("allreduce_max",
           [](Comm &c, py::array_t<double, py::array::c_style |
                                               py::array::forcecast> a) {
             py::gil_scoped_release ng;
             c.allreduce_max(a.mutable_data(), (int64_t)a.size());
           })
      .def("allgather",
           [](Comm &c, int64_t v) {
             py::gil_scoped_release ng;
             return c.allgather(v);
           })
      .def("alltoallv",
           [](Comm &c, py::array_t<double, py::array::c_style |
                                               py::array::forcecast> send,
              const std::vector<int64_t> &counts) {
             std::vector<double> out;
             {
               py::gil_scoped_release ng;
               out = c.alltoallv(send.data(), counts);
             }
             return vec_to_np(std::move(out));
           })
      // device-pointer variants for torch-tensor callers (GPU memory)
      .def("allreduce_sum_device",
           [](Comm &c, uintptr_t ptr, int64_t n) {
             py::gil_scoped_release ng;
             c.allreduce_sum_device((double *)ptr, n);
           })
      // device all-to-all-v: returns (device pointer, total doubles);
      // the buffer is comm-owned and valid until the next device call
      .def("alltoallv_device",
           [](Comm &c, uintptr_t d_send,
              const std::vector<int64_t> &send_counts,
              const std::vector<int64_t> &recv_counts) {
             double *d_recv = nullptr;
             int64_t tot = 0;
             {
               py::gil_scoped_release ng;
               tot = c.alltoallv_device((const double *)d_send, send_counts,
                                        recv_counts, &d_recv);
             }
             return py::make_tuple((uintptr_t)d_recv, tot);
           });
  // debug/test helper: copy n doubles from a device pointer to numpy
  m.def("d2h_doubles", [](uintptr_t ptr, int64_t n) {
    auto out = py::array_t<double>(n);
    if (n) {
      py::gil_scoped_release ng;
      if (hipMemcpy(out.mutable_data(), (const void *)ptr, n * 8,
                    hipMemcpyDeviceToHost) != hipSuccess)
        throw std::runtime_error("d2h_doubles: hipMemcpy failed");
    }
    return out;
  });
  m.def(
      "make_native_comm",
      [](bool want_gpu, int device) {
        auto c = make_comm_from_env(want_gpu, device);
        return c ? c.release() : nullptr; // nullptr => world 1
      },
      py::arg("want_gpu") = true, py::arg("device") = 0,
      py::return_value_policy::take_ownership);
  // Explicit RcclComm construction (any world incl. 1): lets GPU tests
  // exercise the rcclCommInitRank + collective paths without a multi-GPU
  // box.  Returns None when no HIP device is present.
  m.def(
      "make_rccl_comm",
      [](int rank, int world, const std::string &addr, int port, int device) {
        auto c = make_rccl_comm(rank, world, addr, port, device);
        return c ? c.release() : nullptr;
      },
      py::arg("rank") = 0, py::arg("world") = 1,
      py::arg("addr") = "127.0.0.1", py::arg("port") = 29871,
      py::arg("device") = 0, py::return_value_policy::take_ownership);
  m.def("pinned_array", &pinned_array, py::arg("shape"), py::arg("dtype") = "float64");
  m.def("normalize_flux", [](const Mesh &m_, py::array_t<double, py::array::c_style | py::array::forcecast> f) {
    std::vector<double> flux(f.data(), f.data() + f.size());
    return vec_to_np(normalize_flux(m_, flux));
  });
  m.def("write_tally_vtk",
        [](const std::string &path, const Mesh &m_,
           py::array_t<double, py::array::c_style | py::array::forcecast> f) {
          std::vector<double> flux(f.data(), f.data() + f.size());
          write_tally_vtk(path, m_, flux);
        });

  py::class_<PyEngine>(m, "Engine")
      .def(py::init<const Mesh &, int64_t, const std::string &, int, int>(),
           py::arg("mesh"), py::arg("num_particles"), py::arg("device") = "auto",
           py::arg("ngroups") = 1, py::arg("nscores") = 1)
      .def_property_readonly("ngroups",
                             [](const PyEngine &e) { return e.eng->ngroups; })
      .def_property_readonly("nscores",
                             [](const PyEngine &e) { return e.eng->nscores; })
      .def_property_readonly("num_particles",
                             [](const PyEngine &e) { return e.eng->num_particles(); })
      .def_property_readonly("is_gpu", [](const PyEngine &e) { return e.gpu; })
      .def_property(
          "max_steps", [](const PyEngine &e) { return e.eng->max_steps; },
          [](PyEngine &e, int v) { e.eng->max_steps = v; })
      .def("copy_initial_position",
           [](PyEngine &e, py::array_t<double, py::array::c_style | py::array::forcecast> p) {
             if ((int64_t)p.size() != e.eng->num_particles() * 3)
               throw std::runtime_error("positions must have size 3*num_particles");
             e.eng->copy_initial_position(p.data(), e.eng->num_particles());
           })
      .def("move",
           [](PyEngine &e, py::array_t<double, py::array::c_style> origin,
              py::array_t<double, py::array::c_style> dest,
              py::array_t<int8_t, py::array::c_style> flying,
              py::array_t<double, py::array::c_style> weights,
              py::object groups, py::object responses) {
             const int64_t n = e.eng->num_particles();
             if ((int64_t)origin.size() != n * 3 || (int64_t)dest.size() != n * 3 ||
                 (int64_t)flying.size() != n || (int64_t)weights.size() != n)
               throw std::runtime_error("move: array size mismatch");
             const uint16_t *gp = nullptr;
             py::array_t<uint16_t, py::array::c_style | py::array::forcecast> garr;
             if (!groups.is_none()) {
               garr = groups.cast<py::array_t<uint16_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)garr.size() != n)
                 throw std::runtime_error("move: groups size mismatch");
               gp = garr.data();
             }
             const double *rp = nullptr;
             py::array_t<double, py::array::c_style | py::array::forcecast> rarr;
             if (!responses.is_none()) {
               rarr = responses.cast<py::array_t<double, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)rarr.size() != n * e.eng->nscores)
                 throw std::runtime_error("move: responses size must be n*nscores");
               rp = rarr.data();
             }
             py::gil_scoped_release nogil;
             e.eng->move(origin.data(), dest.data(), flying.data(),
                         weights.data(), n, gp, rp);
           },
           py::arg("origin"), py::arg("dest"), py::arg("flying"),
           py::arg("weights"), py::arg("groups") = py::none(),
           py::arg("responses") = py::none())
      .def("move_continue",
           [](PyEngine &e, py::array_t<double, py::array::c_style> dest,
              py::array_t<int8_t, py::array::c_style> flying,
              py::array_t<double, py::array::c_style> weights,
              py::object groups, py::object responses) {
             const int64_t n = e.eng->num_particles();
             if ((int64_t)dest.size() != n * 3 || (int64_t)flying.size() != n ||
                 (int64_t)weights.size() != n)
               throw std::runtime_error("move_continue: array size mismatch");
             const uint16_t *gp = nullptr;
             py::array_t<uint16_t, py::array::c_style | py::array::forcecast> garr;
             if (!groups.is_none()) {
               garr = groups.cast<py::array_t<uint16_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)garr.size() != n)
                 throw std::runtime_error("move_continue: groups size mismatch");
               gp = garr.data();
             }
             const double *rp = nullptr;
             py::array_t<double, py::array::c_style | py::array::forcecast> rarr;
             if (!responses.is_none()) {
               rarr = responses.cast<py::array_t<double, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)rarr.size() != n * e.eng->nscores)
                 throw std::runtime_error("move_continue: responses size must be n*nscores");
               rp = rarr.data();
             }
             py::gil_scoped_release nogil;
             e.eng->move_continue(dest.data(), flying.data(), weights.data(),
                                  n, gp, rp);
           },
           py::arg("dest"), py::arg("flying"), py::arg("weights"),
           py::arg("groups") = py::none(), py::arg("responses") = py::none())
      .def("move_device",
           // Raw device-pointer entry (integers as returned by
           // torch.Tensor.data_ptr()).  origin_ptr=0 means continue
           // semantics; groups_ptr/responses_ptr=0 means none.  The Python
           // wrapper validates device/dtype/shape.
           [](PyEngine &e, uintptr_t origin_ptr, uintptr_t dest_ptr,
              uintptr_t flying_ptr, uintptr_t weights_ptr,
              uintptr_t groups_ptr, uintptr_t responses_ptr) {
             py::gil_scoped_release nogil;
             e.eng->move_device((const double *)origin_ptr,
                                (const double *)dest_ptr,
                                (const int8_t *)flying_ptr,
                                (const double *)weights_ptr,
                                e.eng->num_particles(),
                                (const uint16_t *)groups_ptr,
                                (const double *)responses_ptr);
           },
           py::arg("origin_ptr"), py::arg("dest_ptr"), py::arg("flying_ptr"),
           py::arg("weights_ptr"), py::arg("groups_ptr") = 0,
           py::arg("responses_ptr") = 0)
      .def("walk_raw",
           [](PyEngine &e, py::array_t<double, py::array::c_style | py::array::forcecast> pos,
              py::array_t<double, py::array::c_style | py::array::forcecast> dest,
              py::array_t<int32_t, py::array::c_style | py::array::forcecast> elem,
              py::array_t<double, py::array::c_style | py::array::forcecast> weights,
              py::object groups_obj, py::object responses_obj,
              py::object in_t_obj, py::object in_prev_obj,
              bool resume) -> py::tuple {
             const int64_t n = (int64_t)elem.size();
             if ((int64_t)pos.size() != n * 3 || (int64_t)dest.size() != n * 3 ||
                 (int64_t)weights.size() != n)
               throw std::runtime_error("walk_raw: size mismatch");
             py::array_t<uint16_t, py::array::c_style | py::array::forcecast> groups;
             const uint16_t *gp = nullptr;
             if (!groups_obj.is_none()) {
               groups = groups_obj.cast<
                   py::array_t<uint16_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)groups.size() != n)
                 throw std::runtime_error("walk_raw: groups size mismatch");
               gp = groups.data();
             }
             py::array_t<double, py::array::c_style | py::array::forcecast> resp;
             const double *rp = nullptr;
             if (!responses_obj.is_none()) {
               resp = responses_obj.cast<
                   py::array_t<double, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)resp.size() != n * e.eng->nscores)
                 throw std::runtime_error("walk_raw: responses size must be n*nscores");
               rp = resp.data();
             }
             py::array_t<double, py::array::c_style | py::array::forcecast> in_t;
             const double *tp = nullptr;
             if (!in_t_obj.is_none()) {
               in_t = in_t_obj.cast<
                   py::array_t<double, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)in_t.size() != n)
                 throw std::runtime_error("walk_raw: in_t size mismatch");
               tp = in_t.data();
             }
             py::array_t<int32_t, py::array::c_style | py::array::forcecast> in_prev;
             const int32_t *pp = nullptr;
             if (!in_prev_obj.is_none()) {
               in_prev = in_prev_obj.cast<
                   py::array_t<int32_t, py::array::c_style | py::array::forcecast>>();
               if ((int64_t)in_prev.size() != n)
                 throw std::runtime_error("walk_raw: in_prev size mismatch");
               pp = in_prev.data();
             }
             auto out_pos = py::array_t<double>({n, (int64_t)3});
             auto out_elem = py::array_t<int32_t>(n);
             auto out_status = py::array_t<int8_t>(n);
             auto out_dest = py::array_t<double>({n, (int64_t)3});
             const bool want_res = resume || tp || pp;
             auto out_o = py::array_t<double>({want_res ? n : 0, (int64_t)3});
             auto out_t = py::array_t<double>(want_res ? n : 0);
             auto out_prev = py::array_t<int32_t>(want_res ? n : 0);
             {
               py::gil_scoped_release nogil;
               e.eng->walk_raw(n, pos.data(), dest.data(), elem.data(),
                               weights.data(), out_pos.mutable_data(),
                               out_elem.mutable_data(), out_status.mutable_data(),
                               gp, rp, out_dest.mutable_data(), tp, pp,
                               want_res ? out_o.mutable_data() : nullptr,
                               want_res ? out_t.mutable_data() : nullptr,
                               want_res ? out_prev.mutable_data() : nullptr);
             }
             // out_dest: the walk's final destination (mutated by
             // reflective/periodic restarts); handoffs must resume
             // toward it, not the original dest.  With resume=True the
             // tuple additionally carries (out_o, out_t, out_prev): the
             // bitwise handoff-resume state (walk.h walk_segment doc);
             // feed them back via in_t/in_prev (with pos = out_o row)
             // so the receiving rank replays the walk's fp decisions.
             if (want_res)
               return py::make_tuple(out_pos, out_elem, out_status, out_dest,
                                     out_o, out_t, out_prev);
             return py::make_tuple(out_pos, out_elem, out_status, out_dest);
           },
           py::arg("pos"), py::arg("dest"), py::arg("elem"),
           py::arg("weights"), py::arg("groups") = py::none(),
           py::arg("responses") = py::none(), py::arg("in_t") = py::none(),
           py::arg("in_prev") = py::none(), py::arg("resume") = false)
      .def("walk_raw_device",
           // Raw device-pointer variant for the device-resident partitioned
           // round loop (pointers as from torch.Tensor.data_ptr()).
           [](PyEngine &e, int64_t n, uintptr_t pos, uintptr_t dest,
              uintptr_t elem, uintptr_t weights, uintptr_t out_pos,
              uintptr_t out_elem, uintptr_t out_status, uintptr_t groups,
              uintptr_t responses, uintptr_t out_dest, uintptr_t in_t,
              uintptr_t in_prev, uintptr_t out_o, uintptr_t out_t,
              uintptr_t out_prev) {
             py::gil_scoped_release nogil;
             e.eng->walk_raw_device(
                 n, (const double *)pos, (const double *)dest,
                 (const int32_t *)elem, (const double *)weights,
                 (double *)out_pos, (int32_t *)out_elem, (int8_t *)out_status,
                 (const uint16_t *)groups, (const double *)responses,
                 (double *)out_dest, (const double *)in_t,
                 (const int32_t *)in_prev, (double *)out_o, (double *)out_t,
                 (int32_t *)out_prev);
           },
           py::arg("n"), py::arg("pos"), py::arg("dest"), py::arg("elem"),
           py::arg("weights"), py::arg("out_pos"), py::arg("out_elem"),
           py::arg("out_status"), py::arg("groups") = 0,
           py::arg("responses") = 0, py::arg("out_dest") = 0,
           py::arg("in_t") = 0, py::arg("in_prev") = 0, py::arg("out_o") = 0,
           py::arg("out_t") = 0, py::arg("out_prev") = 0)
      .def("synchronize", [](PyEngine &e) { py::gil_scoped_release nogil; e.eng->synchronize(); })
      .def("flux", [](const PyEngine &e) { return vec_to_np(e.eng->flux()); })
      .def("elem_ids",
           [](const PyEngine &e) {
             auto v = e.eng->elem_ids();
             auto a = py::array_t<int32_t>(v.size());
             std::memcpy(a.mutable_data(), v.data(), v.size() * sizeof(int32_t));
             return a;
           })
      .def("positions",
           [](const PyEngine &e) {
             auto v = e.eng->positions();
             auto a = py::array_t<double>({(int64_t)(v.size() / 3), (int64_t)3});
             std::memcpy(a.mutable_data(), v.data(), v.size() * sizeof(double));
             return a;
           })
      .def("escaped",
           [](const PyEngine &e) {
             auto v = e.eng->escaped();
             auto a = py::array_t<uint8_t>(v.size());
             std::memcpy(a.mutable_data(), v.data(), v.size());
             return a;
           })
      .def("set_flux",
           [](PyEngine &e, py::array_t<double, py::array::c_style | py::array::forcecast> f) {
             e.eng->set_flux(f.data(), (int64_t)f.size());
           })
      .def("set_particle_state",
           [](PyEngine &e, py::array_t<double, py::array::c_style | py::array::forcecast> pos,
              py::array_t<int32_t, py::array::c_style | py::array::forcecast> elem,
              py::array_t<uint8_t, py::array::c_style | py::array::forcecast> escaped) {
             const int64_t n = e.eng->num_particles();
             if ((int64_t)pos.size() != n * 3 || (int64_t)elem.size() != n ||
                 (int64_t)escaped.size() != n)
               throw std::runtime_error("set_particle_state: size mismatch");
             e.eng->set_particle_state(pos.data(), elem.data(), escaped.data(), n);
           })
      .def("end_batch", [](PyEngine &e) { py::gil_scoped_release ng; e.eng->end_batch(); })
      .def("batch_sum", [](const PyEngine &e) { return vec_to_np(e.eng->batch_sum()); })
      .def("batch_sum_sq", [](const PyEngine &e) { return vec_to_np(e.eng->batch_sum_sq()); })
      .def_property_readonly("num_batches",
                             [](const PyEngine &e) { return e.eng->num_batches(); })
      .def("stats", [](const PyEngine &e) {
        const EngineStats &s = e.eng->stats();
        py::dict d;
        d["lost_particles"] = s.lost_particles;
        d["moves"] = s.moves;
        d["relocated"] = s.relocated;
        d["loose_localizations"] = s.loose_localizations;
        return d;
      })
      // First-K lost-walk capture: (k, 4) array of
      // (particle index, drop x, drop y, drop z); empty when nothing lost.
      .def("lost_records", [](const PyEngine &e) {
        auto v = e.eng->lost_records();
        auto arr = vec_to_np(std::move(v));
        return arr.reshape({(py::ssize_t)(arr.size() / 4), (py::ssize_t)4});
      });

  // The 4-call C++ facade, for API-parity tests from Python.
  py::class_<PumiTally>(m, "PumiTally")
      .def(py::init([](const std::string &mesh, int32_t n) {
        int argc = 0;
        char **argv = nullptr;
        return new PumiTally(mesh, n, argc, argv);
      }))
      .def("copy_initial_position",
           [](PumiTally &t, py::array_t<double, py::array::c_style | py::array::forcecast> p) {
             t.CopyInitialPosition((double *)p.data(), (int32_t)p.size());
           })
      .def("move_to_next_location",
           [](PumiTally &t, py::array_t<double, py::array::c_style> origin,
              py::array_t<double, py::array::c_style> dest,
              py::array_t<int8_t, py::array::c_style> flying,
              py::array_t<double, py::array::c_style> weights) {
             t.MoveToNextLocation((double *)origin.data(), (double *)dest.data(),
                                  (int8_t *)flying.mutable_data(),
                                  (double *)weights.data(), (int32_t)origin.size());
           })
      .def("write_tally_results", &PumiTally::WriteTallyResults)
      .def("tally_times", [](const PumiTally &t) {
        py::dict d;
        d["initialization_time"] = t.InitializationTime();
        d["total_time_to_tally"] = t.TallyTime();
        d["vtk_file_write_time"] = t.WriteTime();
        return d;
      });
}
