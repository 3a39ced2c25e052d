"""Long soak of the device-resident walk with end-to-end conservation check."""
import argparse
import time

import numpy as np

import os as _os, sys as _sys
_sys.path.insert(0, _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
import pumiumtally_amd as pt
from pumiumtally_amd.mesh import box_mesh_with_tets
from pumiumtally_amd.utils import make_box_histories


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--steps", type=int, default=300)
    ap.add_argument("--particles", type=int, default=10_000_000)
    ap.add_argument("--mesh-tets", type=int, default=1_000_000)
    ap.add_argument("--mean-chord", type=float, default=8.0)
    args = ap.parse_args()

    import torch

    mesh, cells = box_mesh_with_tets(args.mesh_tets)
    n = args.particles
    eng = pt.TallyEngine(mesh, n, device="cuda:0")
    p0, p1, fl, w = make_box_histories((1, 1, 1), n, args.mean_chord, cells,
                                       pinned=True, sort=False)
    eng.copy_initial_position(p0.reshape(-1))
    eng.synchronize()
    dev = torch.device("cuda:0")
    te = (torch.from_numpy(np.asarray(p0.reshape(-1))).to(dev),
          torch.from_numpy(np.asarray(p1.reshape(-1))).to(dev))
    tf = torch.from_numpy(np.asarray(fl)).to(dev)
    tw = torch.from_numpy(np.asarray(w)).to(dev)
    torch.cuda.synchronize()
    t0 = time.time()
    for k in range(args.steps):
        eng.move_from_device(te[(k + 1) % 2], tf, tw, sync_torch=False)
    eng.synchronize()
    dt = time.time() - t0
    seg = np.linalg.norm(np.asarray(p1) - np.asarray(p0), axis=1)
    expected = args.steps * float((seg * np.asarray(w)).sum())
    got = float(eng.flux().sum())
    lost = eng.stats()["lost_particles"]
    rel = abs(got - expected) / expected
    print(f"{args.steps} steps in {dt:.1f}s = {n*args.steps/dt/1e6:.0f}M ps/s, "
          f"conservation rel err {rel:.2e}, lost {lost}")
    assert rel < 1e-12 and lost == 0, "SOAK FAILED"


if __name__ == "__main__":
    main()
