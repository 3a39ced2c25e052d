"""Mixed-operation burn-in: sustained load across every engine path with
periodic conservation checks.  Exercises full-API moves, continue moves,
device-resident moves, re-localizations, checkpoint save/load, flux reads
and group tallies in one long loop."""
import argparse
import time

import numpy as np

import pumiumtally_amd as pt
from pumiumtally_amd.mesh import box_mesh_with_tets
from pumiumtally_amd.utils import make_box_histories


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=300)
    ap.add_argument("--particles", type=int, default=4_000_000)
    ap.add_argument("--mesh-tets", type=int, default=1_000_000)
    args = ap.parse_args()

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


def ends(p0, p1, k):
    return (p0 if k % 2 == 0 else p1).reshape(-1)


if __name__ == "__main__":
    main()
