"""In-tree build of the native core (_core.so).

Everything is compiled by hipcc for gfx950 (MI355X) in one shared object:
the C++ mesh core, the serial oracle engine, the HIP walk kernels, the
PumiTally C++ facade and the pybind11 bindings.  hipcc cross-compiles the
device code on CPU-only machines; the resulting .so loads anywhere the ROCm
runtime is installed (GPU paths are runtime-gated).

The .so is built in-tree (pumiumtally_amd/_core.so) so it travels with
repo snapshots; it is git-ignored.
"""
import os
import subprocess
import sys
import sysconfig
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
PKG = REPO / "pumiumtally_amd"
SO_PATH = PKG / "_core.so"

SOURCES = [
    "csrc/core/mesh.cpp",
    "csrc/core/mesh_io.cpp",
    "csrc/core/osh_io.cpp",
    "csrc/core/osh_omegah.cpp",
    "csrc/core/engine_cpu.cpp",
    "csrc/core/partition.cpp",
    "csrc/comm/comm_tcp.cpp",
    "csrc/comm/comm_rccl.hip",
    "csrc/hip/engine_gpu.hip",
    "csrc/hip/partition_engine.hip",
    "csrc/api/PumiTally.cpp",
    "csrc/api/pumitally_c.cpp",
    "csrc/pybind/module.cpp",
]
HEADERS = [
    "csrc/core/geom.h",
    "csrc/core/mesh.h",
    "csrc/core/walk.h",
    "csrc/core/engine.h",
    "csrc/comm/comm.h",
    "csrc/core/partition_engine.h",
    "csrc/api/PumiTally.h",
    "csrc/api/pumitally_c.h",
]

GFX_ARCH = os.environ.get("PUMITALLY_GFX_ARCH", "gfx950")


def _needs_build() -> bool:
    if not SO_PATH.exists():
        return True
    so_mtime = SO_PATH.stat().st_mtime
    for rel in SOURCES + HEADERS + ["pumiumtally_amd/_build.py"]:
        p = REPO / rel
        if not p.exists() or p.stat().st_mtime > so_mtime:
            return True
    return False


def _python_includes():
    import pybind11

    return [pybind11.get_include(), sysconfig.get_paths()["include"]]


def build(force: bool = False, verbose: bool = True) -> Path:
    """Compile the native extension for gfx950; no-op if up to date."""
    if not force and not _needs_build():
        return SO_PATH
    hipcc = os.environ.get("HIPCC", "hipcc")
    inc = [f"-I{p}" for p in _python_includes()]
    cmd = (
        # NOTE: no -ffast-math -- the walk's exactness guarantees (bitwise
        # shared-face plane consistency, tolerance discipline) require IEEE
        # semantics, and the kernel is memory-latency bound, not FLOP bound.
        # -ffp-contract=off: fma contraction would make GPU clip points
        # differ from the CPU oracle by ~1ulp; bitwise CPU==GPU walk parity
        # is a test invariant and the walk is latency-bound, not FLOP-bound.
        [hipcc, f"--offload-arch={GFX_ARCH}", "-O3", "-std=c++17", "-fPIC",
         "-shared", "-ffp-contract=off", "-Wno-unused-result",
         "-parallel-jobs=8"]
        + inc
        + [str(REPO / s) for s in SOURCES]
        # RCCL for the library-held multi-GPU comm (csrc/comm)
        + ["-L/opt/rocm/lib", "-lrccl", "-lz", "-Wl,-rpath,/opt/rocm/lib"]
        + ["-o", str(SO_PATH)]
    )
    if verbose:
        print("[pumiumtally_amd] building native core:", " ".join(cmd), flush=True)
    tmp_out = SO_PATH.with_suffix(".so.tmp")
    cmd[-1] = str(tmp_out)
    try:
        subprocess.run(cmd, check=True, cwd=str(REPO))
    except subprocess.CalledProcessError as e:
        raise RuntimeError(f"native build failed (exit {e.returncode})") from e
    os.replace(tmp_out, SO_PATH)
    return SO_PATH


if __name__ == "__main__":
    build(force="--force" in sys.argv)
