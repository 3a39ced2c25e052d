"""Golden-value engine tests.

These re-encode the reference's white-box integration test
(/root/reference/test/test_pumi_tally_impl_methods.cpp) against our engine:
5 particles on the 6-tet unit cube, exact element paths and hand-computed
track lengths to 1e-8, including the boundary clip and the
escaped-particles-skip-phase-A behavior that the second-move flux values
pin down.
"""
import numpy as np
import pytest

import pumiumtally_amd as pt

NP_ = 5


def make_engine(device="cpu"):
    m = pt.build_box(1, 1, 1)
    return pt.TallyEngine(m, NP_, device=device)


def run_golden_sequence(e):
    """Runs the reference test sequence; asserts all pinned values."""
    # ctor: all particles at centroid of element 0
    pos = e.positions()
    assert np.allclose(pos, [0.5, 0.75, 0.25])

    # initial localization at (0.1, 0.4, 0.5) -> all in element 2, no flux
    init = np.tile([0.1, 0.4, 0.5], (NP_, 1)).ravel()
    e.copy_initial_position(init)
    e.synchronize()
    assert list(e.elem_ids()) == [2] * NP_
    assert np.allclose(e.flux(), 0.0)

    # move 1: toward (1.2, 0.4, 0.5); crosses 2 -> 3 -> 4, exits the box.
    dest = np.tile([1.2, 0.4, 0.5], (NP_, 1)).ravel()
    flying = np.ones(NP_, np.int8)
    w = np.ones(NP_)
    e.move(init, dest, flying, w)
    e.synchronize()
    assert list(e.elem_ids()) == [4] * NP_
    # destination clipped to the boundary at x=1.0, not 1.2
    assert np.allclose(e.positions(), [1.0, 0.4, 0.5], atol=1e-8)
    assert (e.escaped() == 1).all()
    f = e.flux()
    expected1 = [0.0, 0.0, 0.3 * NP_, 0.1 * NP_, 0.5 * NP_, 0.0]
    assert np.allclose(f, expected1, atol=1e-8), f

    # move 2: particles 0 and 2 fly to new destinations with weights 2.0 and
    # 0.5; the others are stopped.  The expected flux pins that escaped
    # particles walk from their CLIPPED positions (phase A does not relocate
    # them) -- reference test values at :361-389.
    nxt = np.tile([1.0, 0.4, 0.5], (NP_, 1))
    nxt[0] = [0.15, 0.05, 0.20]
    nxt[2] = [0.85, 0.05, 0.10]
    fl = np.zeros(NP_, np.int8)
    fl[0] = fl[2] = 1
    ww = np.ones(NP_)
    ww[0] = 2.0
    ww[2] = 0.5
    e.move(init, nxt.ravel(), fl, ww)
    e.synchronize()
    ids = e.elem_ids()
    assert list(ids) == [3, 4, 4, 4, 4]
    f2 = e.flux()
    exp3 = 0.1 * NP_ + 0.08790490988459178 * 2.0
    exp4 = 0.5 * NP_ + 0.879049070406094 * 2.0 + 0.552268050859363 * 0.5
    assert abs(f2[3] - exp3) < 1e-8
    assert abs(f2[4] - exp4) < 1e-8
    assert abs(f2[0]) < 1e-12 and abs(f2[1]) < 1e-12 and abs(f2[5]) < 1e-12
    # flux in element 2 unchanged from move 1 (phase A never tallies)
    assert abs(f2[2] - 0.3 * NP_) < 1e-8
    # committed positions = destinations
    p2 = e.positions()
    assert np.allclose(p2, nxt, atol=1e-12)
    assert e.stats()["lost_particles"] == 0


def test_golden_cpu():
    run_golden_sequence(make_engine("cpu"))


@pytest.mark.gpu
def test_golden_gpu():
    e = make_engine("cuda")
    assert e.is_gpu
    run_golden_sequence(e)


def test_phase_a_relocation():
    """A flying, non-escaped particle whose origin changed is relocated
    without tallying (reincarnated/resampled particle semantics)."""
    e = make_engine("cpu")
    init = np.tile([0.1, 0.4, 0.5], (NP_, 1)).ravel()
    e.copy_initial_position(init)
    # resample particle 0 to element 4's region, then fly a tiny segment
    orig = np.tile([0.1, 0.4, 0.5], (NP_, 1))
    orig[0] = [0.9, 0.4, 0.5]
    dest = orig.copy()
    dest[0] = [0.95, 0.4, 0.5]
    fl = np.zeros(NP_, np.int8)
    fl[0] = 1
    w = np.ones(NP_)
    e.move(orig.ravel(), dest.ravel(), fl, w)
    ids = e.elem_ids()
    assert ids[0] == 4          # relocated across the mesh, then walked
    assert list(ids[1:]) == [2] * (NP_ - 1)
    f = e.flux()
    assert abs(f[4] - 0.05) < 1e-12   # only the dest leg tallies
    assert abs(f[2]) < 1e-15 and abs(f[3]) < 1e-15
    assert e.stats()["relocated"] == 1


def test_nonflying_never_moves_or_tallies():
    e = make_engine("cpu")
    init = np.tile([0.1, 0.4, 0.5], (NP_, 1)).ravel()
    e.copy_initial_position(init)
    dest = np.tile([0.9, 0.4, 0.5], (NP_, 1)).ravel()
    e.move(init, dest, np.zeros(NP_, np.int8), np.ones(NP_))
    assert np.allclose(e.flux(), 0.0)
    assert list(e.elem_ids()) == [2] * NP_
    assert np.allclose(e.positions(), [0.1, 0.4, 0.5])


def test_zero_weight_no_tally():
    e = make_engine("cpu")
    init = np.tile([0.1, 0.4, 0.5], (NP_, 1)).ravel()
    e.copy_initial_position(init)
    dest = np.tile([0.9, 0.4, 0.5], (NP_, 1)).ravel()
    e.move(init, dest, np.ones(NP_, np.int8), np.zeros(NP_))
    assert np.allclose(e.flux(), 0.0)
    assert list(e.elem_ids()) == [4] * NP_  # moved, just not tallied


def test_flux_accumulates_across_moves():
    e = make_engine("cpu")
    init = np.tile([0.2, 0.4, 0.5], (NP_, 1)).ravel()
    e.copy_initial_position(init)
    a = np.tile([0.3, 0.4, 0.5], (NP_, 1)).ravel()
    fl = np.ones(NP_, np.int8)
    w = np.ones(NP_)
    e.move(init, a, fl.copy(), w)
    f1 = e.flux().sum()
    assert abs(f1 - 0.1 * NP_) < 1e-12
    b = np.tile([0.25, 0.4, 0.5], (NP_, 1)).ravel()
    e.move(a, b, np.ones(NP_, np.int8), w)
    assert abs(e.flux().sum() - 0.15 * NP_) < 1e-12


@pytest.mark.gpu
def test_move_continue_gpu_matches_cpu():
    m = pt.build_box(6, 6, 6)
    n = 10000
    rng = np.random.default_rng(21)
    o = rng.uniform(0.02, 0.98, size=(n, 3))
    d = rng.uniform(0.02, 0.98, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    flux = {}
    for dev in ("cpu", "cuda"):
        e = pt.TallyEngine(m, n, device=dev)
        e.copy_initial_position(o.ravel())
        e.move_continue(d.ravel(), np.ones(n, np.int8), w)
        e.synchronize()
        flux[dev] = e.flux()
    assert np.abs(flux["cpu"] - flux["cuda"]).max() < 1e-10


@pytest.mark.gpu
def test_move_from_device_matches_host_path():
    torch = pytest.importorskip("torch")
    if not torch.cuda.is_available():
        pytest.skip("no torch cuda")
    m = pt.build_box(6, 6, 6)
    n = 10000
    rng = np.random.default_rng(33)
    o = rng.uniform(0.02, 0.98, size=(n, 3))
    d = rng.uniform(0.02, 0.98, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)

    host = pt.TallyEngine(m, n, device="cuda")
    host.copy_initial_position(o.ravel())
    host.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    host.synchronize()

    devE = pt.TallyEngine(m, n, device="cuda")
    devE.copy_initial_position(o.ravel())
    dev = torch.device("cuda:0")
    t_d = torch.from_numpy(d.ravel()).to(dev)
    t_f = torch.ones(n, dtype=torch.int8, device=dev)
    t_w = torch.from_numpy(w).to(dev)
    devE.move_from_device(t_d, t_f, t_w)
    devE.synchronize()

    assert np.array_equal(host.elem_ids(), devE.elem_ids())
    assert np.abs(host.flux() - devE.flux()).max() < 1e-10
