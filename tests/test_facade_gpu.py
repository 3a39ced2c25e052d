"""BASELINE config-5 analog: the batched PIMPL interface driven by a CPU
'transport' loop with the tally on the GPU, fluxresult.vtk diffed against
the CPU-oracle run of the identical history set."""
import os
import subprocess
import sys

import numpy as np
import pytest

import pumiumtally_amd as pt


def drive(device, osh, out, n=2000, steps=4):
    env_dev = os.environ.get("PUMITALLY_DEVICE")
    os.environ["PUMITALLY_DEVICE"] = device
    os.environ["PUMITALLY_OUTPUT"] = out
    try:
        tally = pt.PumiTally(osh, n)
        rng = np.random.default_rng(31)
        pos = rng.uniform(0.05, 0.95, size=(n, 3))
        w = rng.uniform(0.25, 1.0, n)
        tally.copy_initial_position(pos.ravel().copy())
        for _ in range(steps):
            step = rng.normal(0, 0.1, size=(n, 3))
            dest = np.abs(pos + step)
            dest = np.where(dest > 1.0, 2.0 - dest, dest)
            dest = np.clip(dest, 1e-6, 1 - 1e-6)
            tally.move_to_next_location(pos.ravel().copy(), dest.ravel().copy(),
                                        np.ones(n, np.int8), w.copy())
            pos = dest
        tally.write_tally_results()
    finally:
        if env_dev is None:
            os.environ.pop("PUMITALLY_DEVICE", None)
        else:
            os.environ["PUMITALLY_DEVICE"] = env_dev


@pytest.mark.gpu
def test_facade_gpu_vtk_matches_cpu(tmp_path):
    m = pt.build_box(6, 6, 6)
    osh = str(tmp_path / "mesh.osh")
    m.write_osh(osh)
    cpu_out = str(tmp_path / "flux_cpu.vtk")
    gpu_out = str(tmp_path / "flux_gpu.vtk")
    drive("cpu", osh, cpu_out)
    drive("0", osh, gpu_out)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    r = subprocess.run([sys.executable, os.path.join(repo, "tools", "vtk_diff.py"),
                        cpu_out, gpu_out, "--rtol", "1e-9"],
                       capture_output=True, text=True)
    assert r.returncode == 0, r.stdout + r.stderr
