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

import pumiumtally_amd as pt

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


def test_per_face_reflective_cpu():
    """Quarter-model style: only the +x boundary face reflects; the other
    five faces stay vacuum."""
    m = pt.build_box(3, 3, 3)
    fid, cen, nor = m.boundary_faces()
    plus_x = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    assert len(plus_x) > 0
    m.set_reflective_faces(plus_x)

    n = 8
    o = np.tile([0.5, 0.4, 0.45], (n, 1))
    e = pt.TallyEngine(m, n, device="cpu")

    # exits +x: reflected back inside, full length tallied, no escape
    d = o.copy(); d[:, 0] = 1.4
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), np.ones(n))
    assert (e.escaped() == 0).all()
    assert abs(e.flux().sum() - 0.9 * n) < 1e-10
    p = e.positions()
    assert np.allclose(p[:, 0], 0.6, atol=1e-9)  # folded: 2.0 - 1.4

    # exits -x: vacuum clip as usual
    e2 = pt.TallyEngine(m, n, device="cpu")
    d2 = o.copy(); d2[:, 0] = -0.4
    e2.copy_initial_position(o.ravel())
    e2.move(o.ravel(), d2.ravel(), np.ones(n, np.int8), np.ones(n))
    assert (e2.escaped() == 1).all()
    assert abs(e2.flux().sum() - 0.5 * n) < 1e-10
    assert np.allclose(e2.positions()[:, 0], 0.0, atol=1e-9)


@pytest.mark.gpu
def test_per_face_reflective_gpu():
    m = pt.build_box(3, 3, 3)
    fid, cen, nor = m.boundary_faces()
    m.set_reflective_faces(fid[np.abs(cen[:, 0] - 1.0) < 1e-12])
    n = 2000
    rng = np.random.default_rng(3)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = o + rng.normal(0, 0.4, size=(n, 3))
    w = rng.uniform(0.2, 1.0, n)
    res = {}
    for dev in ("cpu", "cuda"):
        e = pt.TallyEngine(m, n, device=dev)
        e.copy_initial_position(o.ravel())
        e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
        e.synchronize()
        res[dev] = (e.flux(), e.elem_ids(), e.positions(), e.escaped())
    assert np.array_equal(res["cpu"][1], res["cuda"][1])
    assert np.array_equal(res["cpu"][3], res["cuda"][3])
    assert np.allclose(res["cpu"][2], res["cuda"][2], atol=0, rtol=0)
    assert np.abs(res["cpu"][0] - res["cuda"][0]).max() < 1e-10 * max(
        1.0, np.abs(res["cpu"][0]).max())
