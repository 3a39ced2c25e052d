"""Regression: multithreaded CPU move with nscores > 1.

The threaded move path (n >= 65536) sized its per-thread partial tally
as nelems*ngroups, dropping the score dimension: with nscores > 1 the
scored FluxAdd writes k*ngroups*nelems past the buffer -- heap
corruption (double free / segfault) and lost tallies.  Found by
tools/part_world2_soak.py at 400k particles with 2 scores; partials
are now sized to the full flux shape.

This test runs the smallest threaded configuration (70k particles,
2 scores, 2 groups) in a SUBPROCESS (so a recurrence segfaults the
child, not the test runner) and checks the flux elementwise against
the serial path (PUMITALLY_CPU_THREADS=1), which was never affected.
"""
import os
import subprocess
import sys

import numpy as np

SCRIPT = r"""
import os, sys
import numpy as np
sys.path.insert(0, os.environ["PT_ROOT"])
import pumiumtally_amd as pt

n, G, S = 70_000, 2, 2
mesh = pt.build_box(12, 12, 12, 1.0, 1.0, 1.0)
rng = np.random.default_rng(3)
pos = rng.uniform(0.05, 0.95, (n, 3))
dest = np.clip(pos + rng.normal(0, 0.2, (n, 3)), 0.01, 0.99)
w = rng.uniform(0.2, 1.0, n)
grp = rng.integers(0, G, n).astype(np.uint16)
rsp = rng.uniform(0.5, 2.0, (n, S))
eng = pt.TallyEngine(mesh, n, device="cpu", ngroups=G, nscores=S)
eng.copy_initial_position(pos.ravel())
eng.move(pos.ravel(), dest.ravel(), np.ones(n, np.int8), w,
         groups=grp, responses=rsp)
np.save(os.environ["PT_OUT"], np.asarray(eng.flux()))
print("MOVE_OK")
"""


def _run(tmp_path, threads, tag):
    out = str(tmp_path / f"flux_{tag}.npy")
    env = dict(os.environ)
    env["PT_ROOT"] = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env["PT_OUT"] = out
    if threads is not None:
        env["PUMITALLY_CPU_THREADS"] = str(threads)
    r = subprocess.run([sys.executable, "-c", SCRIPT], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, (threads, r.stdout, r.stderr)
    assert "MOVE_OK" in r.stdout
    return np.load(out)


def test_threaded_scored_move_matches_serial(tmp_path):
    f_threaded = _run(tmp_path, None, "mt")   # hardware thread count
    f_serial = _run(tmp_path, 1, "st")
    assert f_threaded.shape == f_serial.shape
    assert np.allclose(f_threaded, f_serial, rtol=1e-12, atol=1e-12), \
        np.abs(f_threaded - f_serial).max()
    assert f_threaded.sum() > 0
