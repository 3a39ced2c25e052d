"""Randomized GPU-vs-CPU differential fuzz on real hardware.

The hardware analog of tools/fuzz_differential.py: each trial draws a
random box mesh, segment set and feature combination (energy groups,
multi-score responses, vacuum / reflective / periodic-x boundary,
fp32 traversal, stateful partitioned engine) and checks the HIP engine
elementwise against the CPU engine fed identical arrays:

  * TallyEngine(cuda) flux == TallyEngine(cpu) flux (elementwise; both
    walks are fp64 plane-parametric so the only divergence allowed is
    atomic-accumulation ordering, covered by the 1e-11 abs tolerance)
  * PartitionedEngine(cuda, world-1) == plain CPU engine
  * fp32 GPU traversal conserves fp64 totals
  * zero lost particles on both devices

Requires a GPU (exits 0 with a notice otherwise, so it is safe in any
harness).  Usage: python tools/fuzz_gpu_differential.py --trials 400
"""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pumiumtally_amd as pt  # noqa: E402


def one_trial(rng, trial):
    nx, ny, nz = rng.integers(1, 8, 3)
    ex, ey, ez = rng.uniform(0.2, 8.0, 3)
    m = pt.build_box(int(nx), int(ny), int(nz), ex, ey, ez)
    bc = rng.choice(["vacuum", "reflective", "periodic"])
    if bc == "periodic":
        fid, cen, _ = m.boundary_faces()
        hi = fid[np.abs(cen[:, 0] - ex) < 1e-9 * ex]
        lo = fid[np.abs(cen[:, 0]) < 1e-9 * ex]
        m.set_periodic_faces(hi, lo, np.array([-ex, 0.0, 0.0]))
    elif bc == "reflective":
        fid, _, _ = m.boundary_faces()
        m.set_reflective_faces(fid)

    n = int(rng.integers(1, 4000))
    G = int(rng.integers(1, 4))
    S = int(rng.integers(1, 4))
    lo_f, hi_f = 1e-5, 1 - 1e-5
    o = rng.uniform(lo_f, hi_f, (n, 3)) * [ex, ey, ez]
    d = rng.uniform(lo_f, hi_f, (n, 3)) * [ex, ey, ez]
    if bc == "periodic":
        sel = rng.random(n) < 0.4
        d[sel, 0] += rng.uniform(0.1, 1.5, int(sel.sum())) * ex \
            * rng.choice([-1.0, 1.0], int(sel.sum()))
    w = rng.uniform(0.0, 2.0, n)
    g = rng.integers(0, G, n).astype(np.uint16) if G > 1 else None
    r = rng.uniform(0.0, 2.0, (n, S)) if S > 1 else None
    fly = np.ones(n, np.int8)

    ref = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), fly.copy(), w, groups=g, responses=r)
    assert ref.stats()["lost_particles"] == 0, (trial, "cpu lost")
    f = np.asarray(ref.flux()).reshape(S, G, m.nelems)

    gpu = pt.TallyEngine(m, n, device="cuda:0", ngroups=G, nscores=S)
    gpu.copy_initial_position(o.ravel())
    gpu.move(o.ravel(), d.ravel(), fly.copy(), w, groups=g, responses=r)
    assert gpu.stats()["lost_particles"] == 0, (trial, "gpu lost")
    fg = np.asarray(gpu.flux()).reshape(S, G, m.nelems)
    assert np.allclose(fg, f, rtol=1e-11, atol=1e-11), \
        (trial, "gpu-vs-cpu", np.abs(fg - f).max())

    if int(rng.integers(0, 2)):
        pe = pt._core.PartitionedEngine(m, n, device="cuda:0", ngroups=G,
                                        nscores=S)
        pe.localize(o.ravel())
        pe.step(d.ravel(), fly.copy(), w, origin=o.ravel(), groups=g,
                responses=r)
        sf = np.asarray(pe.flux_global()).reshape(S, G, m.nelems)
        assert np.allclose(sf, f, atol=1e-11), \
            (trial, "gpu-stateful", np.abs(sf - f).max())

    if int(rng.integers(0, 2)):
        os.environ["PUMITALLY_WALK"] = "fp32"
        try:
            e32 = pt.TallyEngine(m, n, device="cuda:0", ngroups=G, nscores=S)
            e32.copy_initial_position(o.ravel())
            e32.move(o.ravel(), d.ravel(), fly.copy(), w, groups=g,
                     responses=r)
            assert e32.stats()["lost_particles"] == 0, (trial, "fp32 lost")
            f32 = np.asarray(e32.flux()).reshape(S, G, m.nelems)
            assert np.allclose(f32.sum(), f.sum(), rtol=1e-6), \
                (trial, "fp32", f32.sum(), f.sum())
        finally:
            del os.environ["PUMITALLY_WALK"]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=400)
    ap.add_argument("--seed", type=int, default=0)
    args = ap.parse_args()
    if not pt.have_gpu():
        print("fuzz_gpu_differential: no GPU, nothing to do")
        return
    rng = np.random.default_rng(args.seed)
    t0 = time.time()
    for t in range(args.trials):
        one_trial(rng, t)
        if (t + 1) % 50 == 0:
            print(f"  {t+1}/{args.trials} trials ok "
                  f"({time.time()-t0:.1f}s)", flush=True)
    print(f"fuzz_gpu_differential PASS: {args.trials} trials "
          f"(seed {args.seed}) in {time.time()-t0:.1f}s")


if __name__ == "__main__":
    main()
