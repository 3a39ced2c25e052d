"""Periodic boundary conditions: paired boundary faces with a translation.

The reference supports only vacuum boundaries (ApplyVacuumBC,
/root/reference/src/pumitally/PumiTallyImpl.cpp:256-286); reflective and
periodic BCs are extensions.  A walk that exits through a periodic face
teleports its remaining segment by the pair's translation vector and
resumes in the paired element (walk.h periodic_restart) -- an isometry,
so total tallied track length is conserved exactly.

Oracle: a wrapped walk equals the concatenation of plain sub-walks of the
same sub-segments (vacuum walk up to the boundary + fresh walk from the
translated entry point).
"""
import numpy as np
import pytest

import pumiumtally_amd as pt


def _periodic_x_box(nx=3, ny=3, nz=3):
    m = pt.build_box(nx, ny, nz)
    fid, cen, nor = m.boundary_faces()
    hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
    m.set_periodic_faces(hi, lo, np.array([-1.0, 0.0, 0.0]))
    assert m.has_periodic
    return m


def test_periodic_single_wrap_matches_split_oracle():
    m = _periodic_x_box()
    n = 1
    o = np.array([[0.8, 0.41, 0.57]])
    d = np.array([[1.3, 0.41, 0.57]])
    w = np.array([2.0])

    eng = pt.TallyEngine(m, n, device="cpu")
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    got = eng.flux()
    # conservation: the whole 0.5-long segment stays inside the box
    assert abs(got.sum() - 0.5 * w[0]) < 1e-12

    # oracle: vacuum walk 0.8->1.3 (tallies [0.8, 1.0]) + plain walk of the
    # wrapped remainder 0.0->0.3 on the plain mesh
    plain = pt.build_box(3, 3, 3)
    ref = pt.TallyEngine(plain, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    o2 = np.array([[0.0, 0.41, 0.57]])
    d2 = np.array([[0.3, 0.41, 0.57]])
    ref2 = pt.TallyEngine(plain, n, device="cpu")
    ref2.copy_initial_position(o2.ravel())
    ref2.move(o2.ravel(), d2.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(got, ref.flux() + ref2.flux(), atol=1e-12)
    # final particle position is the translated destination
    p = eng.positions().reshape(-1, 3)
    assert np.allclose(p[0], [0.3, 0.41, 0.57], atol=1e-12)
    # particle did NOT escape (it wrapped)
    assert eng._eng.escaped()[0] == 0


def test_periodic_multi_wrap_conservation():
    m = _periodic_x_box(4, 4, 4)
    n = 200
    rng = np.random.default_rng(3)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    # long x-rays wrapping up to 3 times; y/z stay interior
    d = o + np.column_stack([rng.uniform(1.5, 3.2, n),
                             rng.uniform(-0.02, 0.02, n),
                             rng.uniform(-0.02, 0.02, n)])
    d[:, 1:] = np.clip(d[:, 1:], 0.05, 0.95)
    w = rng.uniform(0.1, 1.0, n)
    seg = np.linalg.norm(d - o, axis=1)

    eng = pt.TallyEngine(m, n, device="cpu")
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    total = eng.flux().sum()
    assert eng.stats()["lost_particles"] == 0
    assert abs(total - (seg * w).sum()) < 1e-10 * (seg * w).sum()


def test_periodic_fp32_traversal(monkeypatch):
    monkeypatch.setenv("PUMITALLY_WALK", "fp32")
    m = _periodic_x_box()
    n = 50
    rng = np.random.default_rng(8)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = o + np.column_stack([rng.uniform(0.5, 1.5, n), np.zeros(n), np.zeros(n)])
    w = rng.uniform(0.5, 1.0, n)
    seg = np.linalg.norm(d - o, axis=1)
    eng = pt.TallyEngine(m, n, device="cpu")
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert abs(eng.flux().sum() - (seg * w).sum()) < 1e-10 * (seg * w).sum()


def test_periodic_validation_errors():
    m = pt.build_box(2, 2, 2)
    fid, cen, nor = m.boundary_faces()
    hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
    # wrong translation: no geometric match
    with pytest.raises(RuntimeError, match="no face in B matches"):
        m.set_periodic_faces(hi, lo, np.array([-0.5, 0.0, 0.0]))
    # non-boundary face
    interior = -1
    for f in range(m.nelems * 4):
        if f not in set(fid.tolist()):
            interior = f
            break
    with pytest.raises(RuntimeError, match="not a .*boundary face"):
        m.set_periodic_faces(np.array([interior]), lo[:1],
                             np.array([-1.0, 0.0, 0.0]))
    # size mismatch
    with pytest.raises(RuntimeError, match="differ in size"):
        m.set_periodic_faces(hi, lo[:-1], np.array([-1.0, 0.0, 0.0]))


def test_periodic_partitioned_world1():
    """Periodic BCs in the stateless partitioned driver (round-1 closed
    the feature out; now cross-part periodic faces carry the pair's
    translation in the exchange record -- SubMesh.foreign_shift).  At
    world 1 all pairs are local restarts; oracle = the replicated
    periodic engine."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = _periodic_x_box()
    n = 60
    rng = np.random.default_rng(7)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = o + rng.normal(0, 0.6, size=(n, 3))  # many segments wrap in x
    d[:, 1:] = np.clip(d[:, 1:], 0.02, 0.98)
    w = rng.uniform(0.5, 1.5, n)

    ptal = PartitionedTally(m, device="cpu")
    ptal.run_segments(o, d, w)
    got = ptal.flux_global()

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(got, ref.flux(), atol=1e-12), \
        np.abs(got - ref.flux()).max()


def test_periodic_partitioned_cross_part_shift():
    """Cross-part periodic handoff: partition the periodic box so the
    x=0 and x=1 boundary elements land on DIFFERENT parts, then verify a
    wrapping walk against the replicated periodic oracle.  Runs both
    part engines in one process (world-1 semantics don't apply: we drive
    the submeshes by hand through walk_raw and route the records
    ourselves, exactly what the world-2 driver does)."""
    from pumiumtally_amd import _core

    m = _periodic_x_box()
    # explicit owners: split by x (element centroid): left half part 0
    cents = np.array([m.centroid(t) for t in range(m.nelems)])
    owners = (cents[:, 0] > 0.5).astype(np.int32)
    subs = [_core.extract_submesh(m, owners, p, 1) for p in range(2)]
    shifts = [np.asarray(s.foreign_shift).reshape(-1, 3) for s in subs]
    # the periodic pair faces must appear as foreign entries with a
    # nonzero shift on at least one side
    assert any(np.abs(sh).max() > 0.5 for sh in shifts if sh.size)

    engines = [pt.TallyEngine(s.local, 1, device="cpu") for s in subs]
    g2l = [np.full(m.nelems, -1, np.int64) for _ in range(2)]
    for p in range(2):
        g2l[p][np.asarray(subs[p].elem_l2g)] = np.arange(
            len(subs[p].elem_l2g))

    n = 40
    rng = np.random.default_rng(3)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = o.copy()
    d[:, 0] += rng.uniform(0.3, 1.4, n)  # all wrap through x=1
    d[:, 1:] = np.clip(d[:, 1:] + rng.normal(0, 0.1, size=(n, 2)),
                       0.02, 0.98)
    w = rng.uniform(0.5, 1.5, n)

    gids = m.locate(o)
    assert (gids >= 0).all()
    work = {p: [] for p in range(2)}
    for i in range(n):
        p = owners[gids[i]]
        work[p].append((o[i], d[i], w[i], g2l[p][gids[i]]))
    for _ in range(64):
        moved = 0
        next_work = {p: [] for p in range(2)}
        for p in range(2):
            if not work[p]:
                continue
            arr = work[p]
            pos = np.array([a[0] for a in arr])
            dst = np.array([a[1] for a in arr])
            wg = np.array([a[2] for a in arr])
            el = np.array([a[3] for a in arr], np.int32)
            out_pos, out_elem, status, out_dest = engines[p].walk_raw(
                pos.ravel(), dst.ravel(), el, wg)
            for j in range(len(arr)):
                if status[j] == 2:
                    k = -(int(out_elem[j]) + 2)
                    tg = int(np.asarray(subs[p].foreign_gid)[k])
                    to = int(np.asarray(subs[p].foreign_owner)[k])
                    sh = shifts[p][k]
                    next_work[to].append(
                        (out_pos[j] + sh, out_dest[j] + sh, wg[j],
                         g2l[to][tg]))
                    moved += 1
        work = next_work
        if moved == 0:
            break
    else:
        raise AssertionError("handoff did not converge")

    got = np.zeros(m.nelems)
    for p in range(2):
        got[np.asarray(subs[p].elem_l2g)] += np.asarray(engines[p].flux())

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(got, ref.flux(), atol=1e-12), \
        np.abs(got - ref.flux()).max()


def test_periodic_two_axis():
    """x and y both periodic: diagonal rays wrap in both axes."""
    m = pt.build_box(3, 3, 3)
    fid, cen, nor = m.boundary_faces()
    xhi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    xlo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
    yhi = fid[np.abs(cen[:, 1] - 1.0) < 1e-12]
    ylo = fid[np.abs(cen[:, 1] - 0.0) < 1e-12]
    m.set_periodic_faces(xhi, xlo, np.array([-1.0, 0.0, 0.0]))
    m.set_periodic_faces(yhi, ylo, np.array([0.0, -1.0, 0.0]))

    n = 100
    rng = np.random.default_rng(17)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = o + np.column_stack([rng.uniform(0.5, 1.8, n),
                             rng.uniform(0.5, 1.8, n),
                             rng.uniform(-0.05, 0.05, n)])
    d[:, 2] = np.clip(d[:, 2], 0.05, 0.95)
    w = rng.uniform(0.1, 1.0, n)
    seg = np.linalg.norm(d - o, axis=1)

    eng = pt.TallyEngine(m, n, device="cpu")
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert eng.stats()["lost_particles"] == 0
    assert abs(eng.flux().sum() - (seg * w).sum()) < 1e-10 * (seg * w).sum()


@pytest.mark.gpu
def test_periodic_gpu_matches_cpu():
    m = _periodic_x_box(5, 5, 5)
    n = 20000
    rng = np.random.default_rng(77)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = o + np.column_stack([rng.uniform(0.5, 2.5, n),
                             rng.uniform(-0.02, 0.02, n),
                             rng.uniform(-0.02, 0.02, n)])
    d[:, 1:] = np.clip(d[:, 1:], 0.05, 0.95)
    w = rng.uniform(0.1, 1.0, n)
    seg = np.linalg.norm(d - o, axis=1)

    cpu = pt.TallyEngine(m, n, device="cpu")
    cpu.copy_initial_position(o.ravel())
    cpu.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)

    gpu = pt.TallyEngine(m, n, device="cuda:0")
    assert gpu.is_gpu
    gpu.copy_initial_position(o.ravel())
    gpu.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    gpu.synchronize()

    assert abs(gpu.flux().sum() - (seg * w).sum()) < 1e-10 * (seg * w).sum()
    assert np.abs(cpu.flux() - gpu.flux()).max() < 1e-9
    assert gpu.stats()["lost_particles"] == 0


def test_periodic_grouped_scored_combined():
    """Combined stress: periodic wrap + energy groups + multi-score in one
    move loop (CPU mini version of tools/burnin.py --extended)."""
    m = _periodic_x_box(4, 4, 4)
    n, G, S = 300, 2, 2
    rng = np.random.default_rng(99)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    p1 = rng.uniform(0.05, 0.95, size=(n, 3))
    wrapsel = rng.random(n) < 0.3
    p1[wrapsel, 0] += rng.uniform(0.05, 0.5, int(wrapsel.sum()))
    w = rng.uniform(0.1, 1.0, n)
    groups = rng.integers(0, G, n).astype(np.uint16)
    resp = np.column_stack([np.ones(n), rng.uniform(0.2, 3.0, n)])
    seg = np.linalg.norm(p1 - p0, axis=1)
    per_move = (seg[:, None] * w[:, None] * resp).sum(axis=0)

    eng = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
    eng.copy_initial_position(p0.ravel())
    moves = 5
    for _ in range(moves):
        eng.move(p0.ravel(), p1.ravel(), np.ones(n, np.int8), w,
                 groups=groups, responses=resp)
    assert eng.stats()["lost_particles"] == 0
    totals = eng.flux().reshape(S, -1).sum(axis=1)
    assert np.allclose(totals, moves * per_move, rtol=1e-12)
    by_group = eng.flux().sum(axis=2)
    assert np.allclose(by_group.sum(axis=1), totals, rtol=1e-12)


def test_periodic_walk_raw():
    """walk_raw on a full periodic mesh wraps like move() (the partitioned
    driver refuses periodic, but direct raw walks are supported)."""
    m = _periodic_x_box(3, 3, 3)
    n = 60
    rng = np.random.default_rng(41)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = o + np.column_stack([rng.uniform(0.5, 1.5, n),
                             np.zeros(n), np.zeros(n)])
    w = rng.uniform(0.1, 1.0, n)
    elem = m.locate(o).astype(np.int32)
    seg = np.linalg.norm(d - o, axis=1)

    eng = pt.TallyEngine(m, 1, device="cpu")
    out_pos, out_elem, status, _ = eng.walk_raw(o.ravel(), d.ravel(), elem, w)
    assert (status == 0).all()  # wrapped walks reach their (translated) dest
    assert abs(eng.flux().sum() - (seg * w).sum()) < 1e-10 * (seg * w).sum()


def test_periodic_many_wraps_not_lost():
    """A long segment wrapping the box more times than max_steps worth of
    element crossings is geometrically valid and must not be dropped: a
    periodic restart resets the per-wrap step budget (walk.h), bounded by
    the kMaxWraps wrap cap."""
    m = _periodic_x_box()
    eng = pt.TallyEngine(m, 1, device="cpu")
    eng.max_steps = 40  # ~13 crossings per unit in x at 3 cells/axis
    wraps = 50          # 50 domain crossings >> one max_steps budget
    o = np.array([[0.5, 0.41, 0.57]])
    d = np.array([[0.5 + wraps, 0.41, 0.57]])
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(1, np.int8), np.ones(1))
    assert eng.stats()["lost_particles"] == 0
    # conservation: the whole segment length stays inside the box
    assert abs(eng.flux().sum() - wraps) < 1e-9
    p = eng.positions().reshape(-1, 3)
    assert np.allclose(p[0], [0.5, 0.41, 0.57], atol=1e-7)
