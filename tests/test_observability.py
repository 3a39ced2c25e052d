"""Failure-path observability: loose-tolerance localization counting and
the first-K lost-particle capture (VERDICT round-1 item 7).

The reference only printfs "Not all particles are found"
(/root/reference/src/pumitally/PumiTallyImpl.cpp:455-458) and has no
counter for tolerance-relaxed localizations at all; here both are
counted in stats() and lost walks additionally record (index, drop
position) so a nonzero count on a real mesh is reproducible.
"""
import numpy as np
import pytest

import pumiumtally_amd as pt


def _cpu_engine(n=4):
    mesh = pt.build_box(3, 3, 3, 1.0, 1.0, 1.0)
    return pt.TallyEngine(mesh, n, device="cpu")


def test_loose_localization_counted():
    eng = _cpu_engine(n=2)
    # strict tol is 1e-10 * bbox diagonal (~1.7e-10); a point 1e-8 outside
    # the boundary fails the strict pass but lands inside the tol*1e4
    # retry window (~1.7e-6), exercising exactly the silent escape hatch
    # the counter was added for.
    pos = np.array([[-1e-8, 0.5, 0.5], [0.5, 0.5, 0.5]], dtype=np.float64)
    eng.copy_initial_position(pos.reshape(-1))
    assert all(e >= 0 for e in eng.elem_ids())
    s = eng.stats()
    assert s["loose_localizations"] == 1

    # relocation path (move phase A) counts too
    origin = np.array([[0.25, 0.25, 0.25], [1.0 + 1e-8, 0.5, 0.5]])
    dest = origin + 0.01
    flying = np.ones(2, dtype=np.int8)
    w = np.ones(2)
    eng.move(origin.reshape(-1), dest.reshape(-1), flying, w)
    assert eng.stats()["loose_localizations"] == 2


def test_strict_localization_not_counted():
    eng = _cpu_engine(n=2)
    pos = np.array([[0.2, 0.3, 0.4], [0.7, 0.6, 0.5]])
    eng.copy_initial_position(pos.reshape(-1))
    assert eng.stats()["loose_localizations"] == 0
    assert eng.lost_records().shape == (0, 4)


def test_lost_records_capture():
    eng = _cpu_engine(n=3)
    eng.max_steps = 1  # force every multi-element walk to be dropped
    pos = np.array([[0.05, 0.05, 0.05], [0.5, 0.5, 0.5], [0.9, 0.9, 0.9]])
    eng.copy_initial_position(pos.reshape(-1))
    origin = pos.copy()
    dest = np.array([[0.95, 0.95, 0.95], [0.5, 0.5, 0.5], [0.1, 0.1, 0.1]])
    flying = np.ones(3, dtype=np.int8)
    w = np.ones(3)
    eng.move(origin.reshape(-1), dest.reshape(-1), flying, w)
    s = eng.stats()
    assert s["lost_particles"] == 2  # particles 0 and 2; 1 stays in-element
    rec = eng.lost_records()
    assert rec.shape == (2, 4)
    assert sorted(int(r[0]) for r in rec) == [0, 2]
    # drop positions lie inside the box (the particle was dropped en route)
    assert np.all(rec[:, 1:] >= 0.0) and np.all(rec[:, 1:] <= 1.0)


@pytest.mark.gpu
def test_lost_and_loose_gpu_matches_cpu():
    mesh = pt.build_box(3, 3, 3, 1.0, 1.0, 1.0)
    for dev in ("cpu", "cuda:0"):
        eng = pt.TallyEngine(mesh, 3, device=dev)
        eng.max_steps = 1
        pos = np.array([[-1e-8, 0.5, 0.5], [0.5, 0.5, 0.5], [0.9, 0.9, 0.9]])
        eng.copy_initial_position(pos.reshape(-1))
        origin = np.maximum(pos, 0.0)
        dest = np.array([[0.95, 0.95, 0.95], [0.5, 0.5, 0.5],
                         [0.1, 0.1, 0.1]])
        flying = np.ones(3, dtype=np.int8)
        eng.move(origin.reshape(-1), dest.reshape(-1), flying, np.ones(3))
        s = eng.stats()
        rec = eng.lost_records()
        if dev == "cpu":
            cpu = (s["lost_particles"], s["loose_localizations"],
                   sorted(int(r[0]) for r in rec))
        else:
            assert (s["lost_particles"], s["loose_localizations"],
                    sorted(int(r[0]) for r in rec)) == cpu
