"""Phase-A (resampled-particle relocation) cost at scale.

Real transport steps relocate the fraction of particles that were
resampled since the last step (reincarnation, PumiTally.h:80-86 of the
reference).  This measures move() with F of the batch carrying a fresh
origin (device grid-localization inside the fused kernel), with exact
conservation checked: expected = sum w * |dest - effective_origin|.
"""
import time

import numpy as np

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import pumiumtally_amd as pt
from pumiumtally_amd.mesh import box_mesh_with_tets
from pumiumtally_amd.utils import make_box_histories


def main():
    mesh, cells = box_mesh_with_tets(1_000_000)
    n = 10_000_000
    steps = 10
    rng = np.random.default_rng(0)
    for frac in (0.0, 0.01, 0.1, 1.0):
        eng = pt.TallyEngine(mesh, n, device="cuda:0")
        p0, p1, fl, w = make_box_histories((1, 1, 1), n, 8.0, cells, pinned=True)
        ends = (np.asarray(p0), np.asarray(p1))
        # two origin variants (parity): current positions with F resampled
        k_res = int(frac * n)
        origins, exp_extra = [], []
        for par in range(2):
            o = pt.pinned_array((n, 3), "float64")
            o[:] = ends[par]
            if k_res:
                idx = rng.choice(n, k_res, replace=False)
                # resample NEAR the destination so walk lengths stay
                # comparable to the base workload; the measured delta then
                # isolates the relocation (device localization) cost.
                d_near = ends[1 - par][idx]
                fresh = np.clip(d_near + rng.normal(0, 0.03, size=(k_res, 3)),
                                1e-4, 1 - 1e-4)
                o[idx] = fresh
                # relocated particles walk fresh->dest instead of pos->dest
                d = ends[1 - par]
                base = np.linalg.norm(d - ends[par], axis=1) * np.asarray(w)
                seg = base.sum() - base[idx].sum() + (
                    np.linalg.norm(d[idx] - fresh, axis=1) * np.asarray(w)[idx]).sum()
            else:
                seg = (np.linalg.norm(ends[1 - par] - ends[par], axis=1)
                       * np.asarray(w)).sum()
            origins.append(o)
            exp_extra.append(float(seg))
        eng.copy_initial_position(p0.reshape(-1))
        eng.synchronize()
        expected = 0.0
        # warm
        for k in range(2):
            eng.move(origins[k % 2].reshape(-1), ends[(k + 1) % 2].reshape(-1),
                     fl, w)
            expected += exp_extra[k % 2]
        eng.synchronize()
        t0 = time.time()
        for k in range(2, 2 + steps):
            eng.move(origins[k % 2].reshape(-1), ends[(k + 1) % 2].reshape(-1),
                     fl, w)
            expected += exp_extra[k % 2]
        eng.synchronize()
        dt = (time.time() - t0) / steps
        got = float(eng.flux().sum())
        rel = abs(got - expected) / expected
        assert eng.stats()["lost_particles"] == 0
        print(f"resample frac {frac:4.2f}: {dt*1e3:6.2f} ms/step "
              f"({n/dt/1e6:4.0f}M ps/s), relocated/step ~{k_res}, "
              f"conservation rel err {rel:.1e}")
        del eng


if __name__ == "__main__":
    main()
