"""Run the dependent-chase latency-ceiling probe (tools/chase.hip)."""
import ctypes
import subprocess
from pathlib import Path

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import pumiumtally_amd  # noqa: F401  (loads the HIP runtime consistently)

HERE = Path(__file__).resolve().parent


def main():
    so = HERE / "chase.so"
    src = HERE / "chase.hip"
    if not so.exists() or so.stat().st_mtime < src.stat().st_mtime:
        subprocess.run(
            ["hipcc", "--offload-arch=gfx950", "-O3", "-shared", "-fPIC",
             str(src), "-o", str(so)], check=True)
    lib = ctypes.CDLL(str(so))
    lib.chase_bench.restype = ctypes.c_double
    lib.chase_bench.argtypes = [ctypes.c_int64, ctypes.c_int, ctypes.c_int,
                                ctypes.c_int, ctypes.c_int]
    lib.chase_walklike_bench.restype = ctypes.c_double
    lib.chase_walklike_bench.argtypes = [ctypes.c_int64, ctypes.c_int,
                                         ctypes.c_int, ctypes.c_int,
                                         ctypes.c_int, ctypes.c_int]
    n = 1_000_000  # same table size as the 1M-tet mesh
    for blocks, threads in ((1024, 256), (2048, 256)):
        rate = lib.chase_bench(n, 64, blocks, threads, 5)
        print(f"chase          {blocks}x{threads}: {rate/1e9:.2f} G hops/s")
    for atom in (0, 1, 2):
        rate = lib.chase_walklike_bench(n, 64, 1024, 256, 5, atom)
        print(f"chase+walkwork 1024x256 atomic={atom} (0=none,1=f64,2=u64fixed): {rate/1e9:.2f} G hops/s")
    lib.chase_walklike2_bench.restype = ctypes.c_double
    lib.chase_walklike2_bench.argtypes = lib.chase_walklike_bench.argtypes
    for atom in (0, 1):
        rate = lib.chase_walklike2_bench(n, 64, 1024, 256, 5, atom)
        print(f"chase+walkwork ILP2 1024x256 atomic={atom}: {rate/1e9:.2f} G hops/s")


if __name__ == "__main__":
    main()
