"""Adversarial walk robustness: grazing rays, points exactly on faces,
edges and vertices -- the degenerate geometry cases the reference
outsources to pumipic_adjacency.tpp (SURVEY.md hard part 1)."""
import numpy as np
import pytest

import pumiumtally_amd as pt


def conservation(e, o, d, w):
    e.copy_initial_position(o.ravel())
    ok = e.elem_ids() >= 0
    e.move(o.ravel(), d.ravel(), np.ones(len(w), np.int8), w)
    stats = e.stats()
    return e.flux().sum(), ok, stats


def test_ray_along_internal_face_plane():
    """Segments lying exactly in the x=y diagonal plane (shared faces of the
    6-tet cells) must still tally their full length."""
    m = pt.build_box(3, 3, 3)
    n = 32
    t = np.linspace(0.05, 0.95, n)
    o = np.stack([t * 0 + 0.1, t * 0 + 0.1, t], axis=1)     # on x=y plane
    d = np.stack([t * 0 + 0.9, t * 0 + 0.9, t], axis=1)     # stay on x=y
    w = np.ones(n)
    e = pt.TallyEngine(m, n, device="cpu")
    total, ok, stats = conservation(e, o, d, w)
    assert ok.all()
    assert stats["lost_particles"] == 0
    expected = np.linalg.norm(d - o, axis=1).sum()
    assert abs(total - expected) < 1e-9 * expected


def test_origin_on_vertices():
    """Origins exactly on mesh vertices localize and walk correctly."""
    m = pt.build_box(4, 4, 4)
    coords = m.coords
    # interior vertices only
    inner = coords[np.all((coords > 0.01) & (coords < 0.99), axis=1)][:20]
    n = len(inner)
    d = np.clip(inner + 0.13, 0.0, 0.97)
    e = pt.TallyEngine(m, n, device="cpu")
    total, ok, stats = conservation(e, inner, d, np.ones(n))
    assert ok.all()
    assert stats["lost_particles"] == 0
    expected = np.linalg.norm(d - inner, axis=1).sum()
    assert abs(total - expected) < 1e-9 * expected


def test_dest_exactly_on_face_points():
    """Destinations on cell-boundary planes (x=0.5 etc.)."""
    m = pt.build_box(2, 2, 2)
    n = 16
    rng = np.random.default_rng(4)
    o = rng.uniform(0.05, 0.45, size=(n, 3))
    d = o.copy()
    d[:, 0] = 0.5  # exactly on the internal grid plane
    e = pt.TallyEngine(m, n, device="cpu")
    total, ok, stats = conservation(e, o, d, np.ones(n))
    assert ok.all()
    assert stats["lost_particles"] == 0
    expected = np.linalg.norm(d - o, axis=1).sum()
    assert abs(total - expected) < 1e-9 * max(expected, 1e-300)


def test_segment_along_mesh_edge():
    """A segment running exactly along a cell edge (x=y=0.5 line)."""
    m = pt.build_box(2, 2, 2)
    e = pt.TallyEngine(m, 1, device="cpu")
    o = np.array([0.5, 0.5, 0.1])
    d = np.array([0.5, 0.5, 0.9])
    total, ok, stats = conservation(e, o[None, :], d[None, :], np.ones(1))
    assert ok.all()
    assert stats["lost_particles"] == 0
    assert abs(total - 0.8) < 1e-9


def test_zero_length_segments():
    m = pt.build_box(2, 2, 2)
    n = 8
    rng = np.random.default_rng(5)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    e = pt.TallyEngine(m, n, device="cpu")
    total, ok, stats = conservation(e, o, o.copy(), np.ones(n))
    assert ok.all()
    assert stats["lost_particles"] == 0
    assert abs(total) < 1e-15
    assert np.allclose(e.positions(), o)


def test_long_thin_mesh_many_crossings():
    """1000-crossing chords across an anisotropic mesh: monotone-t walk
    must not get stuck."""
    m = pt.build_box(200, 2, 2, 200.0, 1.0, 1.0)
    n = 4
    o = np.tile([0.05, 0.45, 0.55], (n, 1))
    d = np.tile([199.95, 0.55, 0.45], (n, 1))
    e = pt.TallyEngine(m, n, device="cpu")
    total, ok, stats = conservation(e, o, d, np.ones(n))
    assert ok.all()
    assert stats["lost_particles"] == 0
    expected = np.linalg.norm(d - o, axis=1).sum()
    assert abs(total - expected) < 1e-8


def test_random_stress_no_lost(seed=123):
    """Dense random segments incl. near-degenerate short ones."""
    m = pt.build_box(5, 5, 5)
    n = 3000
    rng = np.random.default_rng(seed)
    o = rng.uniform(0.001, 0.999, size=(n, 3))
    scale = 10.0 ** rng.uniform(-8, -0.3, n)  # lengths from 1e-8 to 0.5
    dirv = rng.normal(size=(n, 3))
    dirv /= np.linalg.norm(dirv, axis=1, keepdims=True)
    d = np.clip(o + scale[:, None] * dirv, 1e-6, 1 - 1e-6)
    e = pt.TallyEngine(m, n, device="cpu")
    total, ok, stats = conservation(e, o, d, np.ones(n))
    assert ok.all()
    assert stats["lost_particles"] == 0
    expected = np.linalg.norm(d - o, axis=1).sum()
    assert abs(total - expected) < 1e-9 * max(1.0, expected)


def test_nonconvex_mesh_void_clipping():
    """Mesh with an interior void (notch): walks clip at the void's vacuum
    boundary exactly like the outer boundary."""
    full = pt.build_box(4, 4, 4)
    coords = full.coords
    tets = full.tet2vert
    # remove the tets of the central 2x2x2-cell block -> interior void
    centroids = coords[tets].mean(axis=1)
    keep = ~np.all((centroids > 0.25) & (centroids < 0.75), axis=1)
    m = pt.mesh_from_arrays(coords, tets[keep])
    assert m.nelems < full.nelems

    e = pt.TallyEngine(m, 1, device="cpu")
    o = np.array([0.1, 0.4, 0.4])     # left of the void, aligned with it
    d = np.array([0.9, 0.4, 0.4])     # would cross the void
    e.copy_initial_position(o)
    e.move(o, d, np.ones(1, np.int8), np.ones(1))
    assert e.stats()["lost_particles"] == 0
    assert e.escaped()[0] == 1
    p = e.positions()[0]
    assert abs(p[0] - 0.25) < 1e-9    # clipped at the void face
    assert abs(e.flux().sum() - 0.15) < 1e-9  # only the pre-void run tallies


def test_second_batch_after_reinitialization():
    """CopyInitialPosition starts a new batch; flux keeps accumulating
    (reference flow: one write at the very end)."""
    m = pt.build_box(2, 2, 2)
    n = 4
    e = pt.TallyEngine(m, n, device="cpu")
    o1 = np.tile([0.2, 0.3, 0.4], (n, 1))
    d1 = o1 + [0.3, 0, 0]
    e.copy_initial_position(o1.ravel())
    e.move(o1.ravel(), d1.ravel(), np.ones(n, np.int8), np.ones(n))
    f1 = e.flux().sum()
    assert abs(f1 - 0.3 * n) < 1e-12
    # new batch, new source positions
    o2 = np.tile([0.6, 0.7, 0.2], (n, 1))
    d2 = o2 + [0, 0.2, 0]
    e.copy_initial_position(o2.ravel())
    assert (e.escaped() == 0).all()
    e.move(o2.ravel(), d2.ravel(), np.ones(n, np.int8), np.ones(n))
    assert abs(e.flux().sum() - (0.3 + 0.2) * n) < 1e-12


def test_extreme_anisotropy_and_scale():
    """Conservation and localization across 10^7:1 aspect ratios and
    mesh scales from 1e-4 to 1e6 (relative-tolerance discipline: loc_tol
    scales with the bbox diagonal, walk t-tolerance is dimensionless)."""
    rng = np.random.default_rng(7)
    for ex, ey, ez in [(1000.0, 1000.0, 0.01), (1e-3, 1.0, 1e3),
                       (5e4, 2.0, 3.0), (1e-4, 1e-4, 1e-4),
                       (1e6, 1e6, 1e6)]:
        m = pt.build_box(4, 4, 4, ex, ey, ez)
        n = 300
        o = rng.uniform(1e-4, 1 - 1e-4, (n, 3)) * [ex, ey, ez]
        d = rng.uniform(1e-4, 1 - 1e-4, (n, 3)) * [ex, ey, ez]
        w = rng.uniform(0.1, 1.0, n)
        e = pt.TallyEngine(m, n, device="cpu")
        e.copy_initial_position(o.ravel())
        assert (np.asarray(e.elem_ids()) >= 0).all(), (ex, ey, ez)
        e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
        expected = (np.linalg.norm(d - o, axis=1) * w).sum()
        assert e.stats()["lost_particles"] == 0, (ex, ey, ez)
        assert abs(e.flux().sum() - expected) < 1e-9 * expected, (ex, ey, ez)
