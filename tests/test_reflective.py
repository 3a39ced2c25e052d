"""Reflective (specular) boundary condition: PUMITALLY_BC=reflective.

An extension beyond the reference (vacuum-only); default reproduces
the reference.  The key invariants: total tallied length equals the FULL
segment length (nothing escapes), and a reflective wall is equivalent to
folding destinations into the box (method of images).
"""
import os
import subprocess
import sys

import numpy as np
import pytest

CODE = r"""
import numpy as np
import pumiumtally_amd as pt

dev = "{dev}"
m = pt.build_box(4, 4, 4)
n = 200
rng = np.random.default_rng(12)
o = rng.uniform(0.1, 0.9, size=(n, 3))
# destinations OUTSIDE the box: with reflective BC the walk folds back in
d = o + rng.normal(0, 0.6, size=(n, 3))
w = rng.uniform(0.2, 1.0, n)

e = pt.TallyEngine(m, n, device=dev)
e.copy_initial_position(o.ravel())
e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
e.synchronize()
assert e.stats()["lost_particles"] == 0
assert (e.escaped() == 0).all()          # nothing leaves a reflective box
total = e.flux().sum()
expected = float((np.linalg.norm(d - o, axis=1) * w).sum())
rel = abs(total - expected) / expected
assert rel < 1e-12, rel                  # full length tallied

# method of images: the final positions equal the folded destinations
def fold(x):
    x = np.abs(x)
    x = np.where(x > 2.0, x - 2.0 * np.floor(x / 2.0), x)
    return np.where(x > 1.0, 2.0 - x, x)

p = e.positions()
assert np.allclose(p, fold(d), atol=1e-9), np.abs(p - fold(d)).max()
print("REFLECT_OK", dev)
"""


def run_case(dev):
    env = dict(os.environ, PUMITALLY_BC="reflective",
               PYTHONPATH=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    r = subprocess.run([sys.executable, "-c", CODE.format(dev=dev)], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0 and f"REFLECT_OK {dev}" in r.stdout, \
        r.stdout + r.stderr


def test_reflective_cpu():
    run_case("cpu")


@pytest.mark.gpu
def test_reflective_gpu():
    run_case("cuda")
