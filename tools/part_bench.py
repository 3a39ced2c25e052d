"""Single-GPU A/B of the partitioned driver's round loop: host-staged
walk_raw vs device-resident walk_raw_device (PUMITALLY_PART_DEVICE).

world=1 has no exchange, so this isolates exactly what the device path
removes: per-round H2D/D2H staging and host-side record building.
    python tools/part_bench.py --segments 2000000 --mesh-tets 1000000
"""
import argparse
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pumiumtally_amd as pt  # noqa: E402
from pumiumtally_amd.mesh import box_mesh_with_tets  # noqa: E402
from pumiumtally_amd.parallel.partition import PartitionedTally  # noqa: E402


def run(mode, mesh, o, d, w, repeats):
    os.environ["PUMITALLY_PART_DEVICE"] = mode
    ptal = PartitionedTally(mesh, device="cuda:0")
    assert ptal._use_device_rounds() == (mode == "1")
    ptal.run_segments(o, d, w)  # warmup
    t0 = time.time()
    for _ in range(repeats):
        ptal.run_segments(o, d, w)
    dt = (time.time() - t0) / repeats
    total = ptal.flux_global().sum()
    return dt, total


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--segments", type=int, default=2_000_000)
    ap.add_argument("--mesh-tets", type=int, default=1_000_000)
    ap.add_argument("--repeats", type=int, default=3)
    args = ap.parse_args()

    mesh, cells = box_mesh_with_tets(args.mesh_tets)
    rng = np.random.default_rng(0)
    n = args.segments
    o = rng.uniform(0.01, 0.99, size=(n, 3))
    d = rng.uniform(0.01, 0.99, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    expected = (np.linalg.norm(d - o, axis=1) * w).sum()

    host_dt, host_total = run("0", mesh, o, d, w, args.repeats)
    dev_dt, dev_total = run("1", mesh, o, d, w, args.repeats)
    for name, tot in (("host", host_total), ("device", dev_total)):
        rel = abs(tot / ((args.repeats + 1) * expected) - 1.0)
        assert rel < 1e-10, (name, rel)
    print(f"host-staged rounds : {host_dt*1e3:8.1f} ms/batch "
          f"({n/host_dt/1e6:.0f}M segments/s)")
    print(f"device-resident    : {dev_dt*1e3:8.1f} ms/batch "
          f"({n/dev_dt/1e6:.0f}M segments/s)  "
          f"speedup {host_dt/dev_dt:.2f}x  [conservation OK]")


if __name__ == "__main__":
    main()
