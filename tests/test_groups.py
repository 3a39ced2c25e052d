"""Energy-group tallies (extension beyond the reference's single scalar
tally; ngroups=1 reproduces reference behavior exactly)."""
import numpy as np
import pytest

import pumiumtally_amd as pt


def run_group_case(device):
    m = pt.build_box(4, 4, 4)
    n = 400
    rng = np.random.default_rng(6)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = rng.uniform(0.05, 0.95, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, 3, n).astype(np.uint16)

    e = pt.TallyEngine(m, n, device=device, ngroups=3)
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g)
    e.synchronize()
    f = e.flux()
    assert f.shape == (3, m.nelems)
    seg = np.linalg.norm(d - o, axis=1) * w
    for grp in range(3):
        expected = seg[g == grp].sum()
        assert abs(f[grp].sum() - expected) < 1e-10 * max(1.0, expected), grp
    return f


def test_groups_cpu():
    run_group_case("cpu")


@pytest.mark.gpu
def test_groups_gpu_matches_cpu():
    fc = run_group_case("cpu")
    fg = run_group_case("cuda")
    assert np.abs(fc - fg).max() < 1e-10 * max(1.0, np.abs(fc).max())


def test_groups_default_is_group0():
    """move() without a groups array tallies into group 0."""
    m = pt.build_box(2, 2, 2)
    n = 10
    o = np.tile([0.2, 0.3, 0.4], (n, 1))
    d = np.tile([0.6, 0.3, 0.4], (n, 1))
    e = pt.TallyEngine(m, n, device="cpu", ngroups=2)
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), np.ones(n))
    f = e.flux()
    assert abs(f[0].sum() - 0.4 * n) < 1e-12
    assert abs(f[1].sum()) < 1e-15


def test_groups_vtk_output(tmp_path):
    m = pt.build_box(2, 2, 2)
    n = 20
    rng = np.random.default_rng(1)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = rng.uniform(0.1, 0.9, size=(n, 3))
    g = (np.arange(n) % 2).astype(np.uint16)
    e = pt.TallyEngine(m, n, device="cpu", ngroups=2)
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), np.ones(n), groups=g)
    out = str(tmp_path / "g.vtk")
    e.write_tally_results(out)
    text = open(out).read()
    assert "SCALARS flux double" in text
    assert "SCALARS flux_g0 double" in text
    assert "SCALARS flux_g1 double" in text


def test_groups_checkpoint_roundtrip(tmp_path):
    m = pt.build_box(2, 2, 2)
    n = 30
    rng = np.random.default_rng(2)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = rng.uniform(0.1, 0.9, size=(n, 3))
    g = rng.integers(0, 2, n).astype(np.uint16)
    e = pt.TallyEngine(m, n, device="cpu", ngroups=2)
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), np.ones(n), groups=g)
    p = str(tmp_path / "c.npz")
    e.save_checkpoint(p)
    e2 = pt.TallyEngine(m, n, device="cpu", ngroups=2)
    e2.load_checkpoint(p)
    assert np.array_equal(e.flux(), e2.flux())
