"""pumiumtally_amd: MI355X-native unstructured-mesh track-length tally engine.

A from-scratch re-implementation of the capabilities of Fuad-HH/PumiUMTally
(PUMI-Tally) for AMD Instinct MI355X: track-length flux tallies on
unstructured tetrahedral meshes for Monte Carlo particle transport, with the
element walk as a hand-written CDNA4 HIP kernel, fp64 throughout, and
multi-GPU scaling over RCCL (see pumiumtally_amd.parallel).

Public surface:
    Mesh / build_box / read_mesh / read_gmsh / read_osh   mesh core
    TallyEngine                                            engine (CPU or GPU)
    PumiTally                                              4-call C++ API parity facade
    parallel.DistributedTally                              multi-GPU driver
"""
from __future__ import annotations

import os as _os

# ROCm runtime ordering: torch wheels bundle their own libamdhip64.so.7
# (same SONAME as /opt/rocm's).  Whichever loads FIRST owns the process;
# loading ours first and torch later leaves torch's HIP dead ("No HIP GPUs
# are available") and poisons subsequent HIP calls.  Importing torch first
# is safe in both orders of use, so do it here, before _core loads, unless
# explicitly disabled.  (Verified on MI355X: torch-first shares one runtime
# and both torch.cuda and our engine work.)
if _os.environ.get("PUMITALLY_NO_TORCH", "0") != "1":
    try:
        import torch as _torch  # noqa: F401
    except ImportError:
        pass


def _load_core():
    try:
        from . import _core  # type: ignore
        return _core
    except ImportError as e:
        if _os.environ.get("PUMITALLY_AUTOBUILD", "1") == "1":
            from ._build import build
            build()
            from . import _core  # type: ignore
            return _core
        raise ImportError(
            "pumiumtally_amd._core is not built. Run "
            "`python -m pumiumtally_amd._build` first."
        ) from e


_core = _load_core()

Mesh = _core.Mesh
PumiTally = _core.PumiTally
build_box = _core.build_box
read_mesh = _core.read_mesh
read_gmsh = _core.read_gmsh
read_osh = _core.read_osh
mesh_from_arrays = _core.mesh_from_arrays
have_gpu = _core.have_gpu
pinned_array = _core.pinned_array
normalize_flux = _core.normalize_flux
write_tally_vtk = _core.write_tally_vtk


class TallyEngine:
    """Track-length tally engine over a tet mesh.

    device: "auto" (GPU if present, else CPU), "cpu", or "cuda[:N]".
    Semantics follow the reference PumiTally flow; see csrc/core/engine.h.
    """

    def __init__(self, mesh, num_particles: int, device: str = "auto",
                 ngroups: int = 1, nscores: int = 1):
        self._eng = _core.Engine(mesh, num_particles, device, ngroups,
                                 nscores)
        self.mesh = mesh
        self.ngroups = ngroups
        self.nscores = nscores

    @property
    def num_particles(self) -> int:
        return self._eng.num_particles

    @property
    def is_gpu(self) -> bool:
        return self._eng.is_gpu

    @property
    def max_steps(self) -> int:
        return self._eng.max_steps

    @max_steps.setter
    def max_steps(self, v: int) -> None:
        self._eng.max_steps = v

    def copy_initial_position(self, positions):
        self._eng.copy_initial_position(positions)

    def move(self, origin, dest, flying, weights, groups=None,
             responses=None):
        """One transport step.  groups (optional): per-particle energy-group
        indices (uint16, in [0, ngroups)); contributions land in
        flux()[group, elem].  responses (optional, n x nscores float64):
        per-particle score multipliers -- score k tallies
        seg*weight*responses[i,k] (flux + heating + ... from one walk).
        The reference has a single scalar tally; ngroups=nscores=1
        (default) matches it exactly."""
        self._eng.move(origin, dest, flying, weights, groups, responses)

    def move_continue(self, dest, flying, weights, groups=None,
                      responses=None):
        """move() without the phase-A origin upload: valid when no particle
        was resampled this step (origin == committed position)."""
        self._eng.move_continue(dest, flying, weights, groups, responses)

    def move_from_device(self, dest, flying, weights, origin=None,
                         sync_torch=True, groups=None, responses=None):
        """Device-resident move: dest/flying/weights (and optionally origin,
        groups, responses) are GPU tensors (torch CUDA tensors or anything
        with data_ptr() semantics via __cuda_array_interface__) already on
        this engine's device -- no host staging.  For GPU-side transport
        codes."""
        def ptr(t, dtype, numel):
            if t is None:
                return 0
            iface = getattr(t, "__cuda_array_interface__", None)
            if iface is None:
                raise TypeError("move_from_device expects CUDA tensors/arrays")
            import math
            n = math.prod(iface["shape"]) if iface["shape"] else 1
            if iface["typestr"] != dtype or n != numel:
                raise TypeError(
                    f"expected {dtype} x{numel}, got {iface['typestr']} x{n}")
            strides = iface.get("strides")
            if strides is not None:
                # C-contiguity required (None == C-contiguous per protocol)
                itemsize = int(dtype[-1])
                expect = []
                acc = itemsize
                for s in reversed(iface["shape"]):
                    expect.append(acc)
                    acc *= s
                if list(strides) != list(reversed(expect)):
                    raise TypeError("move_from_device requires contiguous tensors")
            return iface["data"][0]

        n = self.num_particles
        import sys
        if sync_torch and "torch" in sys.modules:
            # order against torch's stream: tensors produced by torch ops
            # (.to(device), fills) must be materialized before our kernel,
            # which runs on the engine's own HIP stream.
            torch = sys.modules["torch"]
            if torch.cuda.is_available():
                torch.cuda.synchronize()
        self._eng.move_device(
            ptr(origin, "<f8", n * 3), ptr(dest, "<f8", n * 3),
            ptr(flying, "|i1", n), ptr(weights, "<f8", n),
            ptr(groups, "<u2", n), ptr(responses, "<f8", n * self.nscores))

    def walk_raw(self, pos, dest, elem, weights, groups=None,
                 responses=None, in_t=None, in_prev=None, resume=False):
        """Batched raw segment walk (domain-decomposition support): returns
        (out_pos, out_elem, status, out_dest) with status 0=done 1=escaped
        2=handoff 3=lost; tallies into this engine's flux.  out_dest is the
        walk's final destination -- reflective/periodic restarts mutate it,
        and a handoff must resume toward out_dest, not the original dest.
        groups: optional uint16 per-segment energy-group indices (flux row
        group*nelems+elem); responses: optional n x nscores score
        multipliers.  With resume=True (or in_t/in_prev given) the tuple
        additionally carries (out_o, out_t, out_prev) -- the bitwise
        handoff-resume state (csrc/core/walk.h): ship them in the handoff
        record (pos column = out_o row) and seed the receiving walk via
        in_t/in_prev so it replays the sender's fp decisions exactly."""
        return self._eng.walk_raw(pos, dest, elem, weights, groups,
                                  responses, in_t, in_prev, resume)

    def synchronize(self):
        self._eng.synchronize()

    def flux(self):
        """Raw tally.  Shape: (nelems,) for ngroups=nscores=1;
        (ngroups, nelems) for grouped single-score;
        (nscores, nelems) for scored single-group;
        (nscores, ngroups, nelems) for both."""
        f = self._eng.flux()
        if self.nscores > 1 and self.ngroups > 1:
            return f.reshape(self.nscores, self.ngroups, self.mesh.nelems)
        if self.nscores > 1:
            return f.reshape(self.nscores, self.mesh.nelems)
        if self.ngroups > 1:
            return f.reshape(self.ngroups, self.mesh.nelems)
        return f

    def set_flux(self, flux):
        self._eng.set_flux(flux)

    def elem_ids(self):
        return self._eng.elem_ids()

    def positions(self):
        return self._eng.positions()

    def escaped(self):
        return self._eng.escaped()

    def stats(self):
        return self._eng.stats()

    def lost_records(self):
        """First-K records of walks dropped at max_steps: (k, 4) array of
        (particle index, drop x, drop y, drop z).  Pairs with
        stats()["lost_particles"] so a nonzero lost count on a real mesh is
        reproducible, not just counted (the reference only printfs)."""
        return self._eng.lost_records()

    def end_batch(self):
        """Close the current batch: accumulate the per-batch tally into
        running sum / sum-of-squares and zero it (standard MC batch
        statistics; the reference has no variance accounting)."""
        self._eng.end_batch()

    def batch_statistics(self):
        """Returns (mean, rel_std_error) over the closed batches, shaped
        like flux().  rel_std_error = sigma_of_mean / |mean| (0 where
        mean == 0)."""
        import numpy as np

        nb = self._eng.num_batches
        if nb < 1:
            raise RuntimeError("no batches closed yet (call end_batch)")
        s1 = self._eng.batch_sum()
        s2 = self._eng.batch_sum_sq()
        mean = s1 / nb
        var = np.maximum(s2 / nb - mean * mean, 0.0)
        sem = np.sqrt(var / max(nb - 1, 1))
        rel = np.divide(sem, np.abs(mean), out=np.zeros_like(sem),
                        where=mean != 0)
        if self.nscores > 1 and self.ngroups > 1:
            shape = (self.nscores, self.ngroups, self.mesh.nelems)
        elif self.nscores > 1:
            shape = (self.nscores, self.mesh.nelems)
        elif self.ngroups > 1:
            shape = (self.ngroups, self.mesh.nelems)
        else:
            return mean, rel
        return mean.reshape(shape), rel.reshape(shape)

    def save_checkpoint(self, path: str):
        """Persist the full tally state (flux accumulator + particle
        positions/elements/escaped flags) so a crashed batch can resume.
        The reference has no checkpointing (a crash loses the batch)."""
        import numpy as np

        self.synchronize()
        np.savez_compressed(
            path,
            flux=self.flux(),
            positions=self.positions(),
            elem_ids=self.elem_ids(),
            escaped=self.escaped(),
            num_particles=np.int64(self.num_particles),
            nelems=np.int64(self.mesh.nelems),
        )

    def load_checkpoint(self, path: str):
        import numpy as np

        d = np.load(path)
        if int(d["num_particles"]) != self.num_particles or \
           int(d["nelems"]) != self.mesh.nelems:
            raise ValueError("checkpoint does not match engine shape")
        self._eng.set_flux(d["flux"])
        self._eng.set_particle_state(
            np.ascontiguousarray(d["positions"], dtype=np.float64).ravel(),
            np.ascontiguousarray(d["elem_ids"], dtype=np.int32),
            np.ascontiguousarray(d["escaped"], dtype=np.uint8))

    def normalized_flux(self):
        """flux / element volume; shape matches flux() (per group/score
        when ngroups/nscores > 1)."""
        import numpy as np

        f = self.flux()
        if self.nscores > 1 or self.ngroups > 1:
            flat = np.asarray(f).reshape(-1, self.mesh.nelems)
            out = np.stack([_core.normalize_flux(self.mesh, row)
                            for row in flat])
            return out.reshape(np.asarray(f).shape)
        return _core.normalize_flux(self.mesh, f)

    def write_tally_results(self, filename: str = "fluxresult.vtk"):
        if self.nscores > 1:
            # one field per score (score 0 keeps the name "flux"),
            # plus per-group fields when also grouped
            f = self.flux().reshape(self.nscores, self.ngroups,
                                    self.mesh.nelems)
            fields = []
            for k in range(self.nscores):
                name = "flux" if k == 0 else f"score{k}"
                fields.append(
                    (name, _core.normalize_flux(self.mesh, f[k].sum(axis=0))))
                if self.ngroups > 1:
                    fields += [(f"{name}_g{g}",
                                _core.normalize_flux(self.mesh, f[k, g]))
                               for g in range(self.ngroups)]
            self.mesh.write_vtk_fields(filename, fields)
            return
        if self.ngroups > 1:
            # one normalized field per energy group + the total
            f = self.flux()
            fields = [("flux", _core.normalize_flux(self.mesh, f.sum(axis=0)))]
            fields += [(f"flux_g{g}", _core.normalize_flux(self.mesh, f[g]))
                       for g in range(self.ngroups)]
            self.mesh.write_vtk_fields(filename, fields)
            return
        _core.write_tally_vtk(filename, self.mesh, self._eng.flux())


__all__ = [
    "Mesh",
    "PumiTally",
    "TallyEngine",
    "build_box",
    "read_mesh",
    "read_gmsh",
    "read_osh",
    "mesh_from_arrays",
    "have_gpu",
    "pinned_array",
    "normalize_flux",
    "write_tally_vtk",
]
