"""Origin-churn stress for the stateful partitioned engine.

Per step, a fraction of particles is resampled to new origins (the
OpenMC reincarnation pattern) while the rest continue -- exercising the
phase-A relocation, ghost-reroute and host-eject paths repeatedly at
scale.  Oracle: the replicated TallyEngine on the same device fed the
identical streams; flux must match elementwise every N steps.

Usage: python tools/part_churn.py [--particles 2000000] [--steps 50]
       [--resample 0.1] [--mesh-tets 200000] [--device auto]
"""
import argparse
import sys
import time

import numpy as np

sys.path.insert(0, __file__.rsplit("/", 2)[0])
import pumiumtally_amd as pt  # noqa: E402
from pumiumtally_amd.mesh import box_mesh_with_tets  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--particles", type=int, default=2_000_000)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--resample", type=float, default=0.1)
    ap.add_argument("--mesh-tets", type=int, default=200_000)
    ap.add_argument("--device", default="auto")
    ap.add_argument("--check-every", type=int, default=10)
    args = ap.parse_args()

    dev = args.device
    if dev == "auto":
        dev = "cuda:0" if pt.have_gpu() else "cpu"
    if dev == "cpu":
        args.particles = min(args.particles, 50_000)

    mesh, cells = box_mesh_with_tets(args.mesh_tets, extent=1.0)
    n = args.particles
    rng = np.random.default_rng(1)
    pos = rng.uniform(0.05, 0.95, size=(n, 3))

    pe = pt._core.PartitionedEngine(mesh, n, device=dev)
    eng = pt.TallyEngine(mesh, n, device=dev)
    pe.localize(pos.ravel())
    eng.copy_initial_position(pos.ravel())

    t0 = time.time()
    for s in range(args.steps):
        dest = np.clip(pos + rng.normal(0, 0.15, size=(n, 3)), -0.05, 1.05)
        fly = (rng.random(n) > 0.05).astype(np.int8)
        w = rng.uniform(0.1, 1.0, n)
        res = rng.random(n) < args.resample
        origin = pos.copy()
        origin[res] = rng.uniform(-0.02, 1.02, size=(int(res.sum()), 3))
        pe.step(dest.ravel(), fly, w, origin=origin.ravel())
        eng.move(origin.ravel(), dest.ravel(), fly.copy(), w)
        # track like a host app would (approximation is fine: it only
        # shapes the NEXT inputs, both engines see identical arrays)
        pos = np.where(fly[:, None] == 1, np.clip(dest, 0.0, 1.0), origin)
        if (s + 1) % args.check_every == 0 or s + 1 == args.steps:
            f1 = np.asarray(pe.flux_global())
            f2 = np.asarray(eng.flux())
            err = np.abs(f1 - f2).max() / max(f2.max(), 1e-30)
            st = pe.stats()
            print(f"step {s+1:4d}: rel err {err:.3e}, resident {pe.resident}, "
                  f"relocated {st['relocated']}, lost {st['lost_particles']}",
                  flush=True)
            assert err < 1e-9, "DIVERGED"
    dt = time.time() - t0
    print(f"part_churn PASS: {args.steps} steps x {n} particles "
          f"({args.resample:.0%} resampled/step) in {dt:.1f}s on {dev}")


if __name__ == "__main__":
    main()
