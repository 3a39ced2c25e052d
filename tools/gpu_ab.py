"""GPU measurement harness (run on the MI355X box via gpurun).

Measures: H2D link bandwidth from registered memory, sorted-vs-unsorted
particle ordering, full-API vs continue-mode step time, and per-chord walk
scaling.  Results inform the perf notes committed under profiles/.
"""
import time

import numpy as np

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import pumiumtally_amd as pt
from pumiumtally_amd.mesh import box_mesh_with_tets
from pumiumtally_amd.utils import make_box_histories


def time_steps(eng, ends, flying, weights, nsteps=10, warm=3, continue_mode=False):
    for k in range(warm):
        if continue_mode:
            eng.move_continue(ends[(k + 1) % 2], flying, weights)
        else:
            eng.move(ends[k % 2], ends[(k + 1) % 2], flying, weights)
    eng.synchronize()
    t0 = time.time()
    for k in range(warm, warm + nsteps):
        if continue_mode:
            eng.move_continue(ends[(k + 1) % 2], flying, weights)
        else:
            eng.move(ends[k % 2], ends[(k + 1) % 2], flying, weights)
    eng.synchronize()
    return (time.time() - t0) / nsteps


def h2d_bandwidth():
    """Pinned-host H2D link rate via raw hipMemcpy (the staging ceiling)."""
    import ctypes
    n = 256 * 1024 * 1024  # 256 MB
    src = pt.pinned_array((n,), "uint8")
    src[:] = 1
    hip = ctypes.CDLL("libamdhip64.so")
    dptr = ctypes.c_void_p()
    assert hip.hipMalloc(ctypes.byref(dptr), ctypes.c_size_t(n)) == 0
    p_src = src.ctypes.data_as(ctypes.c_void_p)
    # warm
    assert hip.hipMemcpy(dptr, p_src, ctypes.c_size_t(n), 1) == 0  # 1 = H2D
    t0 = time.time()
    reps = 8
    for _ in range(reps):
        hip.hipMemcpy(dptr, p_src, ctypes.c_size_t(n), 1)
    dt = (time.time() - t0) / reps
    hip.hipFree(dptr)
    print(f"[h2d] pinned H2D bandwidth: {n/dt/1e9:.1f} GB/s ({n/1e6:.0f} MB in {dt*1e3:.2f} ms)")


def main():
    h2d_bandwidth()
    mesh, cells = box_mesh_with_tets(1_000_000)
    n = 10_000_000
    for sort in (False, True):
        eng = pt.TallyEngine(mesh, n, device="cuda:0")
        p0, p1, fl, w = make_box_histories((1, 1, 1), n, 8.0, cells, pinned=True, sort=sort)
        eng.copy_initial_position(p0.reshape(-1))
        eng.synchronize()
        ends = (p0.reshape(-1), p1.reshape(-1))
        dt_full = time_steps(eng, ends, fl, w)
        dt_cont = time_steps(eng, ends, fl, w, continue_mode=True)
        lost = eng.stats()["lost_particles"]
        print(f"[ab] sort={sort}: full-API {dt_full*1e3:.2f} ms/step ({n/dt_full/1e6:.0f}M ps/s), "
              f"continue {dt_cont*1e3:.2f} ms/step ({n/dt_cont/1e6:.0f}M ps/s), lost={lost}")
        del eng

    # chord scaling (sorted, continue-mode isolates the walk)
    for chord in (2.0, 8.0, 32.0):
        eng = pt.TallyEngine(mesh, n, device="cuda:0")
        p0, p1, fl, w = make_box_histories((1, 1, 1), n, chord, cells, pinned=True)
        eng.copy_initial_position(p0.reshape(-1))
        eng.synchronize()
        ends = (p0.reshape(-1), p1.reshape(-1))
        dt = time_steps(eng, ends, fl, w, continue_mode=True)
        print(f"[chord] mean_chord={chord}: continue {dt*1e3:.2f} ms/step ({n/dt/1e6:.0f}M ps/s)")
        del eng


if __name__ == "__main__":
    main()
