"""Mixed-operation burn-in: sustained load across every engine path with
periodic conservation checks.  Exercises full-API moves, continue moves,
device-resident moves, re-localizations, checkpoint save/load, flux reads
and group tallies in one long loop."""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pumiumtally_amd as pt  # noqa: E402
from pumiumtally_amd.mesh import box_mesh_with_tets
from pumiumtally_amd.utils import make_box_histories


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=300)
    ap.add_argument("--particles", type=int, default=4_000_000)
    ap.add_argument("--mesh-tets", type=int, default=1_000_000)
    ap.add_argument("--extended", action="store_true",
                    help="exercise energy groups + multi-score responses + "
                         "periodic-x boundaries with per-score conservation")
    args = ap.parse_args()
    if args.extended:
        return extended(args)

    import torch

    mesh, cells = box_mesh_with_tets(args.mesh_tets)
    n = args.particles
    eng = pt.TallyEngine(mesh, n, device="cuda:0")
    p0, p1, fl, w = make_box_histories((1, 1, 1), n, 8.0, cells, pinned=True,
                                       sort=False)
    seg_sum = float((np.linalg.norm(np.asarray(p1) - np.asarray(p0), axis=1)
                     * np.asarray(w)).sum())
    dev = torch.device("cuda:0")
    te = (torch.from_numpy(np.asarray(p0.reshape(-1))).to(dev),
          torch.from_numpy(np.asarray(p1.reshape(-1))).to(dev))
    tf = torch.from_numpy(np.asarray(fl)).to(dev)
    tw = torch.from_numpy(np.asarray(w)).to(dev)
    torch.cuda.synchronize()

    eng.copy_initial_position(p0.reshape(-1))
    tallied = 0  # moves contributing seg_sum each
    k = 0
    cycles = 0
    t_end = time.time() + args.seconds
    t0 = time.time()
    while time.time() < t_end:
        # 6 full-API + 6 continue + 20 device-resident steps
        for _ in range(6):
            eng.move(ends(p0, p1, k), ends(p0, p1, k + 1), fl, w)
            k += 1
            tallied += 1
        for _ in range(6):
            eng.move_continue(ends(p0, p1, k + 1), fl, w)
            k += 1
            tallied += 1
        for _ in range(20):
            eng.move_from_device(te[(k + 1) % 2], tf, tw, sync_torch=False)
            k += 1
            tallied += 1
        # conservation check + checkpoint round-trip every cycle
        total = eng.flux().sum()
        expected = tallied * seg_sum
        rel = abs(total - expected) / expected
        assert rel < 1e-12, (cycles, rel)
        assert eng.stats()["lost_particles"] == 0
        if cycles % 4 == 3:
            eng.save_checkpoint("/tmp/burnin_ckpt.npz")
            eng.load_checkpoint("/tmp/burnin_ckpt.npz")
        if cycles % 7 == 6:
            # re-localize mid-run (parity: new batch) and reset bookkeeping
            eng.copy_initial_position(ends(p0, p1, k).reshape(-1, 3).ravel())
        cycles += 1
    dt = time.time() - t0
    print(f"burn-in OK: {cycles} cycles, {k} moves, "
          f"{n * k / dt / 1e6:.0f}M ps/s avg, conservation rel err {rel:.2e}")


def extended(args):
    """Grouped + scored + periodic mixed soak with per-score conservation.

    30% of destinations are pushed past the +x wall; the mesh is
    x-periodic, so those tracks wrap and the FULL segment length must be
    tallied (vs clipping under vacuum) -- a strict periodic-path gate."""
    import torch

    mesh, cells = box_mesh_with_tets(args.mesh_tets)
    fid, cen, nor = mesh.boundary_faces()
    hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
    mesh.set_periodic_faces(hi, lo, np.array([-1.0, 0.0, 0.0]))

    n = args.particles
    G, S = 2, 2
    eng = pt.TallyEngine(mesh, n, device="cuda:0", ngroups=G, nscores=S)
    p0, p1, fl, w = make_box_histories((1, 1, 1), n, 8.0, cells, pinned=True,
                                       sort=False)
    p0 = np.asarray(p0).copy()
    p1 = np.asarray(p1).copy()
    rng = np.random.default_rng(5)
    wrap = rng.random(n) < 0.3
    p1[wrap, 0] += rng.uniform(0.05, 0.5, int(wrap.sum()))
    groups = rng.integers(0, G, n).astype(np.uint16)
    resp = np.column_stack([np.ones(n), rng.uniform(0.2, 3.0, n)])
    w = np.asarray(w)
    seg = np.linalg.norm(p1 - p0, axis=1)
    per_move = (seg[:, None] * w[:, None] * resp).sum(axis=0)  # per score

    dev = torch.device("cuda:0")
    te = (torch.from_numpy(p0.reshape(-1)).to(dev),
          torch.from_numpy(p1.reshape(-1)).to(dev))
    tf = torch.from_numpy(np.asarray(fl)).to(dev)
    tw = torch.from_numpy(w).to(dev)
    tg = torch.from_numpy(groups).to(dev)
    tr = torch.from_numpy(np.ascontiguousarray(resp)).to(dev)
    torch.cuda.synchronize()

    eng.copy_initial_position(p0.reshape(-1))
    k = 0
    tallied = 0
    cycles = 0
    t_end = time.time() + args.seconds
    t0 = time.time()
    rel = 0.0
    # Always walk p0 -> p1 (origin given, so phase A relocates everyone
    # back to p0 -- wrapped endpoints can lie outside the box, so a
    # ping-pong would start some walks outside the mesh).
    while time.time() < t_end:
        for _ in range(4):
            eng.move(p0.reshape(-1), p1.reshape(-1), fl, w,
                     groups=groups, responses=resp)
            k += 1
            tallied += 1
        for _ in range(12):
            eng.move_from_device(te[1], tf, tw, origin=te[0],
                                 sync_torch=False, groups=tg, responses=tr)
            k += 1
            tallied += 1
        totals = eng.flux().reshape(S, -1).sum(axis=1)
        expected = tallied * per_move
        rel = float(np.abs(totals / expected - 1.0).max())
        assert rel < 1e-12, (cycles, rel, totals, expected)
        assert eng.stats()["lost_particles"] == 0
        # group slices partition each score exactly
        by_group = eng.flux().sum(axis=2)  # (S, G)
        assert np.allclose(by_group.sum(axis=1), totals, rtol=1e-12)
        cycles += 1
    dt = time.time() - t0
    print(f"extended burn-in OK: {cycles} cycles, {k} moves "
          f"(30% periodic-wrapping, {G} groups x {S} scores), "
          f"{args.particles * k / dt / 1e6:.0f}M ps/s avg, "
          f"conservation rel err {rel:.2e}")


def ends(p0, p1, k):
    return (p0 if k % 2 == 0 else p1).reshape(-1)


if __name__ == "__main__":
    main()
