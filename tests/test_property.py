"""Property-based conservation tests (hypothesis).

For ANY box mesh and ANY finite interior segment set, the tally must
conserve total track length exactly and never lose particles.
"""
import numpy as np
import pytest

hyp = pytest.importorskip("hypothesis")
from hypothesis import given, settings, strategies as st  # noqa: E402

import pumiumtally_amd as pt  # noqa: E402


@settings(max_examples=25, deadline=None)
@given(
    cells=st.tuples(st.integers(1, 6), st.integers(1, 6), st.integers(1, 6)),
    extent=st.tuples(
        st.floats(0.1, 50.0, allow_nan=False),
        st.floats(0.1, 50.0, allow_nan=False),
        st.floats(0.1, 50.0, allow_nan=False),
    ),
    seed=st.integers(0, 2**31 - 1),
    n=st.integers(1, 64),
)
def test_conservation_any_box(cells, extent, seed, n):
    m = pt.build_box(*cells, *extent)
    rng = np.random.default_rng(seed)
    lo = np.array(extent) * 1e-6
    hi = np.array(extent) * (1 - 1e-6)
    o = rng.uniform(lo, hi, size=(n, 3))
    d = rng.uniform(lo, hi, size=(n, 3))
    w = rng.uniform(0.0, 3.0, n)
    e = pt.TallyEngine(m, n, device="cpu")
    e.copy_initial_position(o.ravel())
    assert (e.elem_ids() >= 0).all()
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert e.stats()["lost_particles"] == 0
    expected = float((np.linalg.norm(d - o, axis=1) * w).sum())
    got = float(e.flux().sum())
    assert abs(got - expected) <= 1e-9 * max(1.0, expected)
    assert np.allclose(e.positions(), d)


@settings(max_examples=15, deadline=None)
@given(
    seed=st.integers(0, 2**31 - 1),
    axis=st.integers(0, 2),
    direction=st.integers(0, 1),
)
def test_escape_clipping_any_face(seed, axis, direction):
    """Segments leaving through any of the 6 faces clip exactly on it."""
    m = pt.build_box(3, 3, 3)
    rng = np.random.default_rng(seed)
    n = 16
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = o.copy()
    d[:, axis] = 1.5 if direction else -0.5
    e = pt.TallyEngine(m, n, device="cpu")
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), np.ones(n))
    assert e.stats()["lost_particles"] == 0
    assert (e.escaped() == 1).all()
    p = e.positions()
    wall = 1.0 if direction else 0.0
    assert np.allclose(p[:, axis], wall, atol=1e-9)
    inside = np.abs(wall - o[:, axis]).sum()  # axis-aligned: in-mesh length
    assert abs(e.flux().sum() - inside) < 1e-9 * max(1.0, inside)


@settings(max_examples=15, deadline=None)
@given(
    cells=st.tuples(st.integers(1, 5), st.integers(1, 5), st.integers(1, 5)),
    seed=st.integers(0, 2**31 - 1),
    n=st.integers(1, 48),
    nscores=st.integers(2, 4),
)
def test_scored_conservation_any_box(cells, seed, n, nscores):
    """Every score conserves sum(seg*w*resp_k) independently."""
    m = pt.build_box(*cells)
    rng = np.random.default_rng(seed)
    o = rng.uniform(1e-6, 1 - 1e-6, size=(n, 3))
    d = rng.uniform(1e-6, 1 - 1e-6, size=(n, 3))
    w = rng.uniform(0.0, 3.0, n)
    resp = rng.uniform(0.0, 2.0, size=(n, nscores))
    e = pt.TallyEngine(m, n, device="cpu", nscores=nscores)
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, responses=resp)
    assert e.stats()["lost_particles"] == 0
    seg = np.linalg.norm(d - o, axis=1)
    got = e.flux().sum(axis=1)
    expected = (seg[:, None] * w[:, None] * resp).sum(axis=0)
    assert np.allclose(got, expected, rtol=1e-9, atol=1e-12)


@settings(max_examples=15, deadline=None)
@given(
    cells=st.tuples(st.integers(2, 5), st.integers(2, 5), st.integers(2, 5)),
    seed=st.integers(0, 2**31 - 1),
    n=st.integers(1, 32),
    span=st.floats(0.3, 2.8, allow_nan=False),
)
def test_periodic_conservation_any_box(cells, seed, n, span):
    """x-periodic wrap conserves the full segment length for any box
    resolution and wrap count (segments stay y/z-interior)."""
    m = pt.build_box(*cells)
    fid, cen, nor = m.boundary_faces()
    hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
    m.set_periodic_faces(hi, lo, np.array([-1.0, 0.0, 0.0]))
    rng = np.random.default_rng(seed)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = o.copy()
    d[:, 0] += rng.uniform(0.1, span, n) * rng.choice([-1.0, 1.0], n)
    w = rng.uniform(0.0, 2.0, n)
    e = pt.TallyEngine(m, n, device="cpu")
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert e.stats()["lost_particles"] == 0
    expected = float((np.linalg.norm(d - o, axis=1) * w).sum())
    got = float(e.flux().sum())
    assert abs(got - expected) <= 1e-9 * max(1.0, expected)
