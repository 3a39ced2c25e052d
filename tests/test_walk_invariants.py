"""Property-based walk invariants on larger meshes.

The strongest correctness invariant of a track-length tally: for any
segment, the sum of per-element contributions equals weight * (length of
the segment clipped to the mesh).  For interior segments on a convex mesh
that's exactly weight * |dest - orig|.
"""
import numpy as np
import pytest

import pumiumtally_amd as pt


def random_interior_points(rng, n, lo=0.02, hi=1.98):
    return rng.uniform(lo, hi, size=(n, 3))


def test_track_length_conservation_interior():
    m = pt.build_box(6, 6, 6, 2.0, 2.0, 2.0)
    n = 500
    rng = np.random.default_rng(42)
    e = pt.TallyEngine(m, n, device="cpu")
    o = random_interior_points(rng, n)
    d = random_interior_points(rng, n)
    w = rng.uniform(0.1, 2.0, n)
    e.copy_initial_position(o.ravel())
    assert (e.elem_ids() >= 0).all()
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert e.stats()["lost_particles"] == 0
    expected = (np.linalg.norm(d - o, axis=1) * w).sum()
    total = e.flux().sum()
    assert abs(total - expected) < 1e-9 * max(1.0, expected)
    # all particles reached their destination (interior)
    assert (e.escaped() == 0).all()
    assert np.allclose(e.positions(), d)
    # final elements really contain the destinations
    ids = m.locate(d)
    assert (ids == e.elem_ids()).mean() > 0.999  # dest exactly on a face may differ


def test_track_length_conservation_exiting():
    """Segments that leave the box tally exactly the inside part and clip."""
    m = pt.build_box(5, 5, 5, 1.0, 1.0, 1.0)
    n = 200
    rng = np.random.default_rng(7)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    # destinations outside: push past +x face
    d = o.copy()
    d[:, 0] = 1.5
    w = np.ones(n)
    e = pt.TallyEngine(m, n, device="cpu")
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert e.stats()["lost_particles"] == 0
    inside_len = (1.0 - o[:, 0]).sum()
    assert abs(e.flux().sum() - inside_len) < 1e-9
    assert (e.escaped() == 1).all()
    p = e.positions()
    assert np.allclose(p[:, 0], 1.0, atol=1e-9)
    assert np.allclose(p[:, 1:], o[:, 1:], atol=1e-9)


def test_multi_step_history_conservation():
    """A multi-step random-walk history: total flux == total tallied path."""
    m = pt.build_box(4, 4, 4, 1.0, 1.0, 1.0)
    n = 100
    steps = 10
    rng = np.random.default_rng(3)
    e = pt.TallyEngine(m, n, device="cpu")
    cur = rng.uniform(0.1, 0.9, size=(n, 3))
    e.copy_initial_position(cur.ravel())
    total_expected = 0.0
    alive = np.ones(n, bool)
    for _ in range(steps):
        step = rng.normal(0, 0.15, size=(n, 3))
        dest = cur + step
        e.move(cur.ravel(), dest.ravel(), alive.astype(np.int8), np.ones(n))
        # compute expected clipped lengths on the CPU side
        pos = e.positions()
        esc = e.escaped().astype(bool)
        seg = np.linalg.norm(pos - cur, axis=1)
        total_expected += seg[alive].sum()
        # escaped particles die (vacuum BC); others continue from dest
        alive &= ~esc
        cur = pos.copy()
    assert e.stats()["lost_particles"] == 0
    assert abs(e.flux().sum() - total_expected) < 1e-8


def test_axis_aligned_exact():
    """Hand-computable axis ray through a 2x1x1 box (12 tets)."""
    m = pt.build_box(2, 1, 1, 2.0, 1.0, 1.0)
    e = pt.TallyEngine(m, 1, device="cpu")
    o = np.array([0.1, 0.4, 0.5])
    d = np.array([1.9, 0.4, 0.5])
    e.copy_initial_position(o)
    e.move(o, d, np.ones(1, np.int8), np.ones(1))
    f = e.flux()
    assert abs(f.sum() - 1.8) < 1e-12
    # contributions must all be non-negative and land only in crossed cells
    assert (f >= -1e-15).all()


@pytest.mark.gpu
def test_gpu_matches_cpu_oracle():
    """Differential test: GPU flux == CPU flux to 1e-10 on random histories."""
    m = pt.build_box(8, 8, 8, 1.0, 1.0, 1.0)
    n = 20000
    rng = np.random.default_rng(11)
    o = rng.uniform(0.01, 0.99, size=(n, 3))
    d = o + rng.normal(0, 0.3, size=(n, 3))
    w = rng.uniform(0.0, 1.5, n)
    results = {}
    for dev in ("cpu", "cuda"):
        e = pt.TallyEngine(m, n, device=dev)
        e.copy_initial_position(o.ravel())
        e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
        e.synchronize()
        results[dev] = (e.flux(), e.elem_ids(), e.positions(), e.escaped())
    fc, ic, pc, ec = results["cpu"]
    fg, ig, pg, eg = results["cuda"]
    assert np.array_equal(ic, ig)
    assert np.array_equal(ec, eg)
    assert np.allclose(pc, pg, atol=0, rtol=0)  # walks are bitwise identical
    scale = np.abs(fc).max()
    assert np.abs(fc - fg).max() < 1e-10 * max(scale, 1.0)


@pytest.mark.gpu
def test_gpu_multistep_conservation_long_chords():
    """Back-to-back moves with no intermediate sync at long mean chord:
    regression test for the staging-vs-kernel WAR race across steps."""
    import pumiumtally_amd as ptm
    from pumiumtally_amd.mesh import box_mesh_with_tets
    from pumiumtally_amd.utils import make_box_histories

    mesh, cells = box_mesh_with_tets(200_000)
    n = 2_000_000
    steps = 6
    e = ptm.TallyEngine(mesh, n, device="cuda")
    p0, p1, fl, w = make_box_histories((1, 1, 1), n, 32.0, cells, pinned=True)
    e.copy_initial_position(p0.reshape(-1))
    seg = np.linalg.norm(np.asarray(p1) - np.asarray(p0), axis=1)
    expected_per_step = float((seg * np.asarray(w)).sum())
    ends = (p0.reshape(-1), p1.reshape(-1))
    for k in range(steps):
        e.move(ends[k % 2], ends[(k + 1) % 2], fl, w)
    e.synchronize()
    total = e.flux().sum()
    assert e.stats()["lost_particles"] == 0
    rel = abs(total - steps * expected_per_step) / (steps * expected_per_step)
    assert rel < 1e-12, rel


@pytest.mark.gpu
def test_gpu_matches_cpu_on_irregular_meshes():
    """Differential CPU-vs-GPU on anisotropic and notched (non-convex)
    meshes, not just uniform boxes."""
    rng = np.random.default_rng(17)

    # anisotropic box
    m1 = pt.build_box(20, 5, 3, 4.0, 1.0, 0.3)
    # notched box (interior void)
    full = pt.build_box(6, 6, 6)
    cent = full.coords[full.tet2vert].mean(axis=1)
    keep = ~np.all((cent > 0.33) & (cent < 0.67), axis=1)
    m2 = pt.mesh_from_arrays(full.coords, full.tet2vert[keep])

    for m, box in ((m1, (4.0, 1.0, 0.3)), (m2, (1.0, 1.0, 1.0))):
        n = 8000
        o = rng.uniform(0.01, 0.99, size=(n, 3)) * box
        d = np.clip(o + rng.normal(0, 0.2, size=(n, 3)) * np.array(box),
                    1e-5, np.array(box) - 1e-5)
        w = rng.uniform(0.1, 1.0, n)
        res = {}
        for dev in ("cpu", "cuda"):
            e = pt.TallyEngine(m, n, device=dev)
            e.copy_initial_position(o.ravel())
            e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
            e.synchronize()
            res[dev] = (e.flux(), e.elem_ids(), e.positions())
        assert np.array_equal(res["cpu"][1], res["cuda"][1])
        assert np.allclose(res["cpu"][2], res["cuda"][2], atol=0, rtol=0)
        scale = max(1.0, np.abs(res["cpu"][0]).max())
        assert np.abs(res["cpu"][0] - res["cuda"][0]).max() < 1e-10 * scale


@pytest.mark.gpu
def test_gpu_matches_cpu_across_resorts():
    """40 drifting moves (crossing several periodic device re-sorts of the
    particle slots): per-particle state in caller order must stay bitwise
    equal to the CPU oracle -- pins the s2c permutation plumbing."""
    m = pt.build_box(6, 6, 6)
    n = 20000
    rng = np.random.default_rng(97)
    start = rng.uniform(0.05, 0.95, size=(n, 3))
    segs = []
    cur = start
    for _ in range(40):
        nxt = np.clip(cur + rng.normal(0, 0.06, size=(n, 3)), 0.01, 0.99)
        segs.append(nxt)
        cur = nxt
    w = rng.uniform(0.1, 1.0, n)

    res = {}
    for dev in ("cpu", "cuda"):
        e = pt.TallyEngine(m, n, device=dev)
        e.copy_initial_position(start.ravel())
        prev = start
        for nxt in segs:
            e.move(prev.ravel(), nxt.ravel(), np.ones(n, np.int8), w)
            prev = nxt
        e.synchronize()
        res[dev] = (e.flux(), e.elem_ids(), e.positions(), e.escaped(),
                    e.stats()["lost_particles"])
    assert res["cpu"][4] == 0 and res["cuda"][4] == 0
    assert np.array_equal(res["cpu"][1], res["cuda"][1])
    assert np.array_equal(res["cpu"][3], res["cuda"][3])
    assert np.allclose(res["cpu"][2], res["cuda"][2], atol=0, rtol=0)
    scale = max(1.0, np.abs(res["cpu"][0]).max())
    assert np.abs(res["cpu"][0] - res["cuda"][0]).max() < 1e-9 * scale


@pytest.mark.gpu
def test_gpu_sort_disabled_config():
    """PUMITALLY_SORT=0 (no device sorting) must stay correct."""
    import subprocess
    import sys
    code = (
        "import numpy as np, pumiumtally_amd as pt\n"
        "m = pt.build_box(6, 6, 6)\n"
        "n = 20000\n"
        "rng = np.random.default_rng(11)\n"
        "o = rng.uniform(0.02, 0.98, size=(n, 3))\n"
        "d = np.clip(o + rng.normal(0, 0.2, size=(n, 3)), 1e-5, 1 - 1e-5)\n"
        "w = rng.uniform(0.1, 1.0, n)\n"
        "res = {}\n"
        "for dev in ('cpu', 'cuda'):\n"
        "    a, b = o, d\n"
        "    e = pt.TallyEngine(m, n, device=dev)\n"
        "    e.copy_initial_position(a.ravel())\n"
        "    for _ in range(3):\n"
        "        e.move(a.ravel(), b.ravel(), np.ones(n, np.int8), w)\n"
        "        a, b = b, a\n"
        "    e.synchronize()\n"
        "    res[dev] = (e.flux(), e.elem_ids())\n"
        "assert np.array_equal(res['cpu'][1], res['cuda'][1])\n"
        "assert np.abs(res['cpu'][0] - res['cuda'][0]).max() < 1e-9 * max(1.0, np.abs(res['cpu'][0]).max())\n"
        "print('SORT_OFF_OK')\n"
    )
    import os as _os
    env = dict(_os.environ, PUMITALLY_SORT="0",
               PYTHONPATH=_os.path.dirname(_os.path.dirname(_os.path.abspath(__file__))))
    r = subprocess.run([sys.executable, "-c", code], env=env,
                       capture_output=True, text=True, timeout=300)
    assert r.returncode == 0 and "SORT_OFF_OK" in r.stdout, r.stdout + r.stderr


@pytest.mark.gpu
def test_immediate_move_after_localization():
    """copy_initial_position followed IMMEDIATELY by move (no sync):
    regression test for the first-move staging race on d_origin_[0]."""
    import pumiumtally_amd as ptm
    from pumiumtally_amd.mesh import box_mesh_with_tets

    mesh, _ = box_mesh_with_tets(500_000)
    n = 2_000_000  # large enough that localization takes a while
    rng = np.random.default_rng(41)
    o = rng.uniform(0.01, 0.99, size=(n, 3))
    d = np.clip(o + rng.normal(0, 0.05, size=(n, 3)), 1e-5, 1 - 1e-5)
    w = rng.uniform(0.1, 1.0, n)
    e = ptm.TallyEngine(mesh, n, device="cuda")
    e.copy_initial_position(o.ravel())
    e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)  # no sync between
    e.synchronize()
    expected = float((np.linalg.norm(d - o, axis=1) * w).sum())
    got = float(e.flux().sum())
    assert abs(got - expected) < 1e-10 * expected
    assert e.stats()["lost_particles"] == 0


def test_fp32_traversal_flux_agreement():
    """PUMITALLY_WALK=fp32 vs fp64: totals identical (both tally in fp64),
    per-element flux within sliver tolerance (fp32 only picks exit faces)."""
    import subprocess
    import sys
    import os as _os
    code = (
        "import numpy as np, pumiumtally_amd as pt\n"
        "m = pt.build_box(6, 6, 6)\n"
        "n = 30000\n"
        "rng = np.random.default_rng(5)\n"
        "o = rng.uniform(0.02, 0.98, size=(n, 3))\n"
        "d = np.clip(o + rng.normal(0, 0.15, size=(n, 3)), 1e-5, 1 - 1e-5)\n"
        "w = rng.uniform(0.1, 1.0, n)\n"
        "e = pt.TallyEngine(m, n, device='cpu')\n"
        "e.copy_initial_position(o.ravel())\n"
        "e.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)\n"
        "np.save('/tmp/fp_flux_' + __import__('os').environ['PUMITALLY_WALK'], e.flux())\n"
    )
    repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
    for mode in ("fp64", "fp32"):
        env = dict(_os.environ, PUMITALLY_WALK=mode, PYTHONPATH=repo)
        r = subprocess.run([sys.executable, "-c", code], env=env,
                           capture_output=True, text=True, timeout=300)
        assert r.returncode == 0, r.stderr
    f64 = np.load("/tmp/fp_flux_fp64.npy")
    f32 = np.load("/tmp/fp_flux_fp32.npy")
    assert abs(f64.sum() - f32.sum()) < 1e-9 * f64.sum()
    # per-element: identical up to rare near-tie face-order slivers
    scale = np.abs(f64).max()
    assert np.abs(f64 - f32).max() < 1e-5 * scale
