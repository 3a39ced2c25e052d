"""Randomized differential fuzz across the full feature matrix.

Each trial draws a random box mesh, a random segment set, and a random
feature combination (energy groups, multi-score responses, boundary
condition: vacuum / reflective / periodic-x, fp32 traversal, partitioned
vs plain) and checks every applicable oracle:

  * conservation: total tally == sum(seg_inside * w * resp) (exact for
    interior/periodic; computed per escaping segment for vacuum via the
    walk itself being compared across configurations)
  * partitioned == plain engine (elementwise)
  * scored == per-score reweighted single-score runs (elementwise)
  * fp32 traversal == fp64 totals (to conservation tolerance)
  * zero lost particles everywhere

    python tools/fuzz_differential.py --trials 300 [--seed 1]
"""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pumiumtally_amd as pt  # noqa: E402


def one_trial(rng, trial):
    nx, ny, nz = rng.integers(1, 7, 3)
    ex, ey, ez = rng.uniform(0.2, 8.0, 3)
    m = pt.build_box(int(nx), int(ny), int(nz), ex, ey, ez)
    bc = rng.choice(["vacuum", "reflective", "periodic"])
    if bc == "periodic":
        fid, cen, nor = m.boundary_faces()
        hi = fid[np.abs(cen[:, 0] - ex) < 1e-9 * ex]
        lo = fid[np.abs(cen[:, 0]) < 1e-9 * ex]
        m.set_periodic_faces(hi, lo, np.array([-ex, 0.0, 0.0]))
    elif bc == "reflective":
        fid, cen, nor = m.boundary_faces()
        m.set_reflective_faces(fid)

    n = int(rng.integers(1, 120))
    G = int(rng.integers(1, 4))
    S = int(rng.integers(1, 4))
    lo_f, hi_f = 1e-5, 1 - 1e-5
    o = rng.uniform(lo_f, hi_f, (n, 3)) * [ex, ey, ez]
    d = rng.uniform(lo_f, hi_f, (n, 3)) * [ex, ey, ez]
    if bc == "periodic":
        # push some x destinations outside so tracks wrap
        sel = rng.random(n) < 0.4
        d[sel, 0] += rng.uniform(0.1, 1.5, int(sel.sum())) * ex \
            * rng.choice([-1.0, 1.0], int(sel.sum()))
    w = rng.uniform(0.0, 2.0, n)
    g = rng.integers(0, G, n).astype(np.uint16) if G > 1 else None
    r = rng.uniform(0.0, 2.0, (n, S)) if S > 1 else None
    seg = np.linalg.norm(d - o, axis=1)

    eng = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)
    assert eng.stats()["lost_particles"] == 0, (trial, "lost")
    f = np.asarray(eng.flux()).reshape(S, G, m.nelems)

    if bc in ("reflective", "periodic"):
        # isometry BCs conserve the full segment set exactly
        resp = r if r is not None else np.ones((n, 1))
        expected = (seg[:, None] * w[:, None] * resp).sum(axis=0)
        got = f.sum(axis=(1, 2))
        ok = np.allclose(got, expected, rtol=1e-9, atol=1e-12)
        assert ok, (trial, bc, got, expected)

    # scored == reweighted single-score
    if S > 1:
        k = int(rng.integers(0, S))
        ref = pt.TallyEngine(m, n, device="cpu", ngroups=G)
        ref.copy_initial_position(o.ravel())
        ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w * r[:, k],
                 groups=g)
        rf = np.asarray(ref.flux()).reshape(G, m.nelems)
        assert np.allclose(f[k], rf, rtol=1e-9, atol=1e-12), (trial, "score")

    # partitioned == plain (since round 2 this includes periodic: the
    # exchange record carries the pair translation + walk destination)
    if int(rng.integers(0, 2)):
        from pumiumtally_amd.parallel.partition import PartitionedTally
        ptal = PartitionedTally(m, device="cpu", ngroups=G, nscores=S,
                                ghost_rings=int(rng.integers(0, 2)))
        ptal.run_segments(o, d, w, groups=g, responses=r)
        pf = np.asarray(ptal.flux_global()).reshape(S, G, m.nelems)
        assert np.allclose(pf, f, atol=1e-11), (trial, "partitioned",
                                                np.abs(pf - f).max())

    # stateful PartitionedEngine == plain (world-1; multi-step with the
    # same segments replayed through localize+step; periodic is world-1
    # legal there too)
    if int(rng.integers(0, 2)):
        pe = pt._core.PartitionedEngine(m, n, device="cpu", ngroups=G,
                                        nscores=S)
        pe.localize(o.ravel())
        pe.step(d.ravel(), np.ones(n, np.int8), w, origin=o.ravel(),
                groups=g, responses=r)
        sf = np.asarray(pe.flux_global()).reshape(S, G, m.nelems)
        assert np.allclose(sf, f, atol=1e-11), (trial, "stateful",
                                                np.abs(sf - f).max())

    # fp32 traversal conserves the same totals
    if int(rng.integers(0, 2)):
        os.environ["PUMITALLY_WALK"] = "fp32"
        try:
            e32 = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
            e32.copy_initial_position(o.ravel())
            e32.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g,
                     responses=r)
            assert e32.stats()["lost_particles"] == 0, (trial, "fp32 lost")
            f32 = np.asarray(e32.flux()).reshape(S, G, m.nelems)
            assert np.allclose(f32.sum(), f.sum(), rtol=1e-9), (trial, "fp32")
        finally:
            del os.environ["PUMITALLY_WALK"]


def one_snapped_trial(rng, trial):
    """Adversarial geometry: origins/destinations snapped exactly onto
    mesh vertices, edge midpoints and face centroids.  Under an isometry
    BC (reflective everywhere) total track length must STILL be conserved
    exactly and nothing may be lost -- this probes the walk's tolerance
    discipline at measure-zero configurations."""
    nx, ny, nz = rng.integers(1, 5, 3)
    m = pt.build_box(int(nx), int(ny), int(nz))
    fid, cen, nor = m.boundary_faces()
    m.set_reflective_faces(fid)
    verts = np.asarray(m.coords).reshape(-1, 3)
    n = int(rng.integers(4, 64))

    def snapped_points(k):
        kind = rng.integers(0, 3, k)
        pts = np.empty((k, 3))
        vi = rng.integers(0, len(verts), k)
        vj = rng.integers(0, len(verts), k)
        vk = rng.integers(0, len(verts), k)
        pts[kind == 0] = verts[vi[kind == 0]]                       # vertex
        pts[kind == 1] = 0.5 * (verts[vi[kind == 1]] + verts[vj[kind == 1]])
        pts[kind == 2] = (verts[vi[kind == 2]] + verts[vj[kind == 2]]
                          + verts[vk[kind == 2]]) / 3.0
        return pts

    o = snapped_points(n)
    d = snapped_points(n)
    w = rng.uniform(0.1, 2.0, n)
    seg = np.linalg.norm(d - o, axis=1)

    eng = pt.TallyEngine(m, n, device="cpu")
    eng.copy_initial_position(o.ravel())
    assert (np.asarray(eng.elem_ids()) >= 0).all(), (trial, "snap locate")
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert eng.stats()["lost_particles"] == 0, (trial, "snap lost")
    expected = (seg * w).sum()
    got = float(np.asarray(eng.flux()).sum())
    assert abs(got - expected) <= 1e-9 * max(1.0, expected), \
        (trial, "snap conservation", got, expected)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--trials", type=int, default=300)
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--snapped", action="store_true",
                    help="adversarial on-vertex/edge/face geometry instead")
    args = ap.parse_args()
    rng = np.random.default_rng(args.seed)
    t0 = time.time()
    for trial in range(args.trials):
        (one_snapped_trial if args.snapped else one_trial)(rng, trial)
        if (trial + 1) % 50 == 0:
            print(f"{trial + 1}/{args.trials} trials OK "
                  f"({time.time() - t0:.0f}s)", flush=True)
    print(f"fuzz_differential{' (snapped)' if args.snapped else ''}: "
          f"{args.trials} trials PASS in {time.time() - t0:.0f}s")


if __name__ == "__main__":
    main()
