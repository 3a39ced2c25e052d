"""Stateful partitioned engine (csrc/hip/partition_engine.hip) vs the
replicated single-engine oracle.

Residency semantics under test (matching Engine::move, engine.h):
non-flying particles stay put; escaped particles keep clipped pos/elem
and are never phase-A relocated; resampled origins relocate (possibly
across ranks); out-of-mesh particles tally nothing and track their
requested position.  Flux equality vs the oracle is the full-system
check: any residency/handoff/localization bug shows up as a missing or
double-counted track segment.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

import pumiumtally_amd as pt

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _histories(mesh, n, seed, steps):
    """Per-step (origin, dest, flying, weights) with resamples/stops."""
    rng = np.random.default_rng(seed)
    out = []
    pos = rng.uniform(0.05, 0.95, size=(n, 3))
    for s in range(steps):
        dest = np.clip(pos + rng.normal(0, 0.25, size=(n, 3)), -0.2, 1.2)
        flying = (rng.random(n) > 0.15).astype(np.int8)
        w = rng.uniform(0.1, 1.0, n)
        # resample ~10% of particles to brand-new origins
        res = rng.random(n) < 0.10
        origin = pos.copy()
        origin[res] = rng.uniform(-0.05, 1.05, size=(int(res.sum()), 3))
        out.append((origin, dest, flying, w))
        # committed position approximation for the NEXT step's origin: the
        # oracle tracks the true one; we just need plausible inputs, so
        # use dest clipped into the box for flying particles
        pos = np.where(flying[:, None] == 1, np.clip(dest, 0.0, 1.0), origin)
    return out


def _run_oracle(mesh, n, hist):
    eng = pt.TallyEngine(mesh, n, device="cpu")
    eng.copy_initial_position(hist[0][0].ravel())
    for origin, dest, flying, w in hist:
        eng.move(origin.ravel(), dest.ravel(), flying.copy(), w)
    return eng


def test_partition_engine_world1_matches_oracle():
    mesh = pt.build_box(4, 4, 4)
    n = 300
    hist = _histories(mesh, n, seed=11, steps=5)

    pe = pt._core.PartitionedEngine(mesh, n, device="cpu")
    assert pe.world == 1
    pe.localize(hist[0][0].ravel())
    for origin, dest, flying, w in hist:
        pe.step(dest.ravel(), flying, w, origin=origin.ravel())
    got = pe.flux_global()

    eng = _run_oracle(mesh, n, hist)
    want = eng.flux()
    assert np.allclose(got, want, atol=1e-12), np.abs(got - want).max()

    # per-particle state parity (positions + escaped bookkeeping): at
    # world 1 every particle is resident here
    assert pe.resident == n
    assert np.allclose(np.asarray(pe.positions()).ravel(),
                       np.asarray(eng.positions()).ravel(), atol=1e-12)


def test_partition_engine_world1_continue_steps():
    """origin=None (continue) steps: no relocation scans at all."""
    mesh = pt.build_box(4, 4, 4)
    n = 200
    rng = np.random.default_rng(4)
    p0 = rng.uniform(0.1, 0.9, size=(n, 3))
    p1 = rng.uniform(0.1, 0.9, size=(n, 3))
    w = rng.uniform(0.5, 1.5, n)
    fly = np.ones(n, np.int8)

    pe = pt._core.PartitionedEngine(mesh, n, device="cpu")
    pe.localize(p0.ravel())
    for k in range(4):
        dest = p1 if k % 2 == 0 else p0
        pe.step(dest.ravel(), fly, w)  # continue semantics

    eng = pt.TallyEngine(mesh, n, device="cpu")
    eng.copy_initial_position(p0.ravel())
    for k in range(4):
        dest = p1 if k % 2 == 0 else p0
        eng.move_continue(dest.ravel(), fly.copy(), w)
    assert np.allclose(pe.flux_global(), eng.flux(), atol=1e-12)
    assert pe.stats()["relocated"] == 0


def test_partition_engine_groups_world1():
    mesh = pt.build_box(3, 3, 3)
    n = 150
    rng = np.random.default_rng(9)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    p1 = rng.uniform(0.05, 0.95, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, 3, n).astype(np.uint16)
    fly = np.ones(n, np.int8)

    pe = pt._core.PartitionedEngine(mesh, n, device="cpu", ngroups=3)
    pe.localize(p0.ravel())
    pe.step(p1.ravel(), fly, w, groups=g)

    eng = pt.TallyEngine(mesh, n, device="cpu", ngroups=3)
    eng.copy_initial_position(p0.ravel())
    eng.move(p0.ravel(), p1.ravel(), fly.copy(), w, groups=g)
    assert np.allclose(pe.flux_global(),
                       np.asarray(eng.flux()).ravel(), atol=1e-12)


WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
device = os.environ.get("PT_DEVICE", "cpu")

mesh = pt.build_box(4, 4, 4)
n = 400
steps = 5
rng = np.random.default_rng(21)  # same stream on all ranks
pos = rng.uniform(0.05, 0.95, size=(n, 3))
hist = []
for s in range(steps):
    dest = np.clip(pos + rng.normal(0, 0.3, size=(n, 3)), -0.1, 1.1)
    flying = (rng.random(n) > 0.1).astype(np.int8)
    w = rng.uniform(0.1, 1.0, n)
    res = rng.random(n) < 0.08
    origin = pos.copy()
    origin[res] = rng.uniform(0.0, 1.0, size=(int(res.sum()), 3))
    hist.append((origin, dest, flying, w))
    pos = np.where(flying[:, None] == 1, np.clip(dest, 0.0, 1.0), origin)

grp = rng.integers(0, 2, n).astype(np.uint16)
rsp = rng.uniform(0.5, 2.0, size=(n, 2))
pe = pt._core.PartitionedEngine(mesh, n, device=device, ngroups=2,
                                nscores=2)
assert pe.world == world, pe.world
pe.localize(hist[0][0].ravel())
for origin, dest, flying, w in hist:
    pe.step(dest.ravel(), flying, w, origin=origin.ravel(), groups=grp,
            responses=rsp)
got = pe.flux_global()

if rank == 0:
    eng = pt.TallyEngine(mesh, n, device="cpu", ngroups=2, nscores=2)
    eng.copy_initial_position(hist[0][0].ravel())
    for origin, dest, flying, w in hist:
        eng.move(origin.ravel(), dest.ravel(), flying.copy(), w,
                 groups=grp, responses=rsp)
    want = np.asarray(eng.flux()).ravel()
    assert np.allclose(got, want, atol=1e-11), np.abs(got - want).max()
    print("PART_ENGINE_WORLD2_OK resident=", pe.resident)
"""


def _spawn2(tmp_path, device):
    script = tmp_path / "w.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.update({
        "WORLD_SIZE": "2",
        "MASTER_ADDR": "127.0.0.1",
        "PUMITALLY_PORT": str(25000 + (os.getpid() + 31) % 15000),
        "PUMITALLY_NO_TORCH": "1",
        "PUMITALLY_COMM": "tcp" if device != "cpu" else "",
        "PT_DEVICE": device,
        "PYTHONPATH": ROOT,
    })
    if not env["PUMITALLY_COMM"]:
        del env["PUMITALLY_COMM"]
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = "0"
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=300)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PART_ENGINE_WORLD2_OK" in outs[0]


def test_partition_engine_world2_cpu(tmp_path):
    """Real 2-part decomposition with cross-rank handoffs and cross-rank
    resampling, against the single-engine oracle."""
    _spawn2(tmp_path, "cpu")


@pytest.mark.gpu
def test_partition_engine_world1_gpu_matches_oracle():
    mesh = pt.build_box(4, 4, 4)
    n = 300
    hist = _histories(mesh, n, seed=11, steps=5)
    pe = pt._core.PartitionedEngine(mesh, n, device="cuda:0")
    pe.localize(hist[0][0].ravel())
    for origin, dest, flying, w in hist:
        pe.step(dest.ravel(), flying, w, origin=origin.ravel())
    got = pe.flux_global()
    eng = _run_oracle(mesh, n, hist)
    assert np.allclose(got, eng.flux(), atol=1e-11)
    assert pe.resident == n
    assert np.allclose(np.asarray(pe.positions()).ravel(),
                       np.asarray(eng.positions()).ravel(), atol=1e-12)


@pytest.mark.gpu
def test_partition_engine_world2_gpu_shared(tmp_path):
    """Two ranks share GPU 0 (TCP comm): the whole device pipeline --
    prepare/gather/walk/collect/pack/unpack kernels + host-staged
    exchange -- runs on real hardware with real handoffs."""
    _spawn2(tmp_path, "cuda:0")


def test_partition_engine_scored_world1():
    """nscores + responses through the stateful engine: responses are
    gathered by gid like dest/weights (never shipped in records)."""
    mesh = pt.build_box(3, 3, 3)
    n = 120
    rng = np.random.default_rng(17)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    p1 = rng.uniform(0.05, 0.95, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    resp = rng.uniform(0.5, 2.0, size=(n, 2))
    fly = np.ones(n, np.int8)

    pe = pt._core.PartitionedEngine(mesh, n, device="cpu", nscores=2)
    pe.localize(p0.ravel())
    pe.step(p1.ravel(), fly, w, responses=resp)

    eng = pt.TallyEngine(mesh, n, device="cpu", nscores=2)
    eng.copy_initial_position(p0.ravel())
    eng.move(p0.ravel(), p1.ravel(), fly.copy(), w, responses=resp)
    assert np.allclose(pe.flux_global(),
                       np.asarray(eng.flux()).ravel(), atol=1e-12)


def test_step_local_world1_matches_step():
    """Coupled-host path at world 1: resident_list + frame-ordered
    step_local must reproduce the global-array step exactly (same
    segments, same flux, same committed state)."""
    mesh = pt.build_box(4, 4, 4)
    n = 250
    rng = np.random.default_rng(31)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    steps = []
    pos = p0.copy()
    for _ in range(3):
        d = np.clip(pos + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
        w = rng.uniform(0.1, 1.0, n)
        fly = (rng.random(n) > 0.1).astype(np.int8)
        steps.append((d, fly, w))
        pos = np.where(fly[:, None] == 1, d, pos)

    a = pt._core.PartitionedEngine(mesh, n, device="cpu", ngroups=2,
                                   nscores=2)
    b = pt._core.PartitionedEngine(mesh, n, device="cpu", ngroups=2,
                                   nscores=2)
    grp = rng.integers(0, 2, n).astype(np.uint16)
    rsp = rng.uniform(0.5, 2.0, size=(n, 2))
    a.localize(p0.ravel())
    b.localize(p0.ravel())
    for d, fly, w in steps:
        a.step(d.ravel(), fly, w, groups=grp, responses=rsp)
        gids = b.resident_list()
        # world 1: all particles resident, frame order == gid order
        b.step_local(d[gids].ravel(), fly[gids], w[gids],
                     groups=grp[gids], responses=rsp[gids])
    fa, fb = a.flux_global(), b.flux_global()
    assert np.allclose(fa, fb, atol=1e-12), np.abs(fa - fb).max()
    assert np.allclose(a.positions(), b.positions(), atol=1e-15)


LOCAL_WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
device = os.environ.get("PT_DEVICE", "cpu")

mesh = pt.build_box(4, 4, 4)
n = 400
steps = 4
rng = np.random.default_rng(47)  # same stream on all ranks
p0 = rng.uniform(0.05, 0.95, size=(n, 3))
hist = []
pos = p0.copy()
for _ in range(steps):
    d = np.clip(pos + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
    fly = (rng.random(n) > 0.1).astype(np.int8)
    w = rng.uniform(0.1, 1.0, n)
    hist.append((d, fly, w))
    pos = np.where(fly[:, None] == 1, d, pos)

pe = pt._core.PartitionedEngine(mesh, n, device=device)
pe.localize(p0.ravel())
for d, fly, w in hist:
    gids = pe.resident_list()
    # each rank feeds ONLY its residents' inputs, in frame order -- the
    # coupled-host contract (nothing global-sized crosses to the engine)
    pe.step_local(d[gids].ravel(), fly[gids], w[gids])
got = pe.flux_global()

if rank == 0:
    ref = pt.TallyEngine(mesh, n, device="cpu")
    ref.copy_initial_position(p0.ravel())
    for d, fly, w in hist:
        ref.move_continue(d.ravel(), fly.copy(), w)
    want = ref.flux()
    assert np.allclose(got, want, atol=1e-11), np.abs(got - want).max()
    print("STEP_LOCAL_WORLD2_OK resident=", pe.resident)
"""


def test_step_local_world2_cpu(tmp_path):
    """Coupled-host world-2: per-rank frame-ordered inputs, records carry
    weight+destination across ranks, flux == replicated oracle."""
    script = tmp_path / "w.py"
    script.write_text(LOCAL_WORKER)
    env = dict(os.environ)
    env.update({
        "WORLD_SIZE": "2",
        "MASTER_ADDR": "127.0.0.1",
        "PUMITALLY_PORT": str(26000 + (os.getpid() + 77) % 14000),
        "PUMITALLY_NO_TORCH": "1",
        "PT_DEVICE": "cpu",
        "PYTHONPATH": ROOT,
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = "0"
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=300)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "STEP_LOCAL_WORLD2_OK" in outs[0]


@pytest.mark.gpu
def test_step_local_world1_gpu():
    mesh = pt.build_box(4, 4, 4)
    n = 300
    rng = np.random.default_rng(53)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    d = np.clip(p0 + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
    w = rng.uniform(0.1, 1.0, n)
    fly = np.ones(n, np.int8)

    pe = pt._core.PartitionedEngine(mesh, n, device="cuda:0")
    pe.localize(p0.ravel())
    gids = pe.resident_list()
    assert len(gids) == n
    pe.step_local(d[gids].ravel(), fly[gids], w[gids])

    ref = pt.TallyEngine(mesh, n, device="cpu")
    ref.copy_initial_position(p0.ravel())
    ref.move_continue(d.ravel(), fly.copy(), w)
    assert np.allclose(pe.flux_global(), ref.flux(), atol=1e-11)


def test_state_transfer_checkpoint_roundtrip():
    """Decomposition-independent snapshot: walk, snapshot, rebuild a
    FRESH engine, restore, continue -- total flux equals the
    uninterrupted run (mid-batch flux is drained host-side)."""
    mesh = pt.build_box(4, 4, 4)
    n = 200
    rng = np.random.default_rng(61)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    d1 = np.clip(p0 + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
    d2 = np.clip(d1 + rng.normal(0, 0.3, size=(n, 3)), -0.1, 1.1)
    w = rng.uniform(0.1, 1.0, n)
    fly = np.ones(n, np.int8)

    # uninterrupted
    a = pt._core.PartitionedEngine(mesh, n, device="cpu")
    a.localize(p0.ravel())
    a.step(d1.ravel(), fly, w)
    a.step(d2.ravel(), fly.copy(), w)
    want = a.flux_global()

    # interrupted after step 1: snapshot -> fresh engine -> restore
    b = pt._core.PartitionedEngine(mesh, n, device="cpu")
    b.localize(p0.ravel())
    b.step(d1.ravel(), fly, w)
    snap = (b.positions().copy(), b.elem_ids_global().copy(),
            b.escaped_mask().copy())
    flux_so_far = np.asarray(b.flux_global()).copy()
    del b
    c = pt._core.PartitionedEngine(mesh, n, device="cpu")
    c.set_state(*snap)
    c.step(d2.ravel(), fly.copy(), w)
    got = flux_so_far + np.asarray(c.flux_global())
    assert np.allclose(got, want, atol=1e-12), np.abs(got - want).max()


def test_repartition_by_reconstruction():
    """Dynamic load balance for the stateful engine: build a NEW engine
    with work-weighted owners and transfer the state; the continued walk
    matches the un-repartitioned run."""
    mesh = pt.build_box(4, 4, 4)
    n = 200
    rng = np.random.default_rng(67)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    d1 = np.clip(p0 + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
    d2 = np.clip(d1 + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
    w = rng.uniform(0.1, 1.0, n)
    fly = np.ones(n, np.int8)

    a = pt._core.PartitionedEngine(mesh, n, device="cpu")
    a.localize(p0.ravel())
    a.step(d1.ravel(), fly, w)
    flux1 = np.asarray(a.flux_global()).copy()
    snap = (a.positions(), a.elem_ids_global(), a.escaped_mask())
    # "measured work" = the first step's flux; weighted Morton owners
    owners = pt._core.partition_morton(mesh, 1, flux1 + 1e-12)
    b = pt._core.PartitionedEngine(mesh, n, device="cpu",
                                   owners=np.asarray(owners, np.int32))
    b.set_state(*snap)
    b.step(d2.ravel(), fly.copy(), w)
    got = flux1 + np.asarray(b.flux_global())

    ref = pt._core.PartitionedEngine(mesh, n, device="cpu")
    ref.localize(p0.ravel())
    ref.step(d1.ravel(), fly.copy(), w)
    ref.step(d2.ravel(), fly.copy(), w)
    want = ref.flux_global()
    assert np.allclose(got, want, atol=1e-12), np.abs(got - want).max()


@pytest.mark.gpu
def test_state_transfer_gpu_matches_cpu():
    mesh = pt.build_box(4, 4, 4)
    n = 200
    rng = np.random.default_rng(71)
    p0 = rng.uniform(0.05, 0.95, size=(n, 3))
    d1 = np.clip(p0 + rng.normal(0, 0.3, size=(n, 3)), -0.05, 1.05)
    d2 = np.clip(d1 + rng.normal(0, 0.3, size=(n, 3)), 0.02, 0.98)
    w = rng.uniform(0.1, 1.0, n)
    fly = np.ones(n, np.int8)

    outs = {}
    for dev in ("cpu", "cuda:0"):
        a = pt._core.PartitionedEngine(mesh, n, device=dev)
        a.localize(p0.ravel())
        a.step(d1.ravel(), fly.copy(), w)
        f1 = np.asarray(a.flux_global()).copy()
        snap = (a.positions(), a.elem_ids_global(), a.escaped_mask())
        b = pt._core.PartitionedEngine(mesh, n, device=dev)
        b.set_state(*snap)
        b.step(d2.ravel(), fly.copy(), w)
        outs[dev] = f1 + np.asarray(b.flux_global())
    assert np.allclose(outs["cpu"], outs["cuda:0"], atol=1e-10)


def test_edge_cases_small_and_idle():
    """Degenerate shapes: 1 particle, all-stopped steps, empty frames,
    step after everyone escaped."""
    mesh = pt.build_box(2, 2, 2)
    pe = pt._core.PartitionedEngine(mesh, 1, device="cpu")
    pe.localize(np.array([0.5, 0.6, 0.7]))
    assert pe.resident == 1
    # all-stopped step: nothing moves, nothing tallies
    pe.step(np.array([0.9, 0.6, 0.7]), np.zeros(1, np.int8), np.ones(1))
    assert pe.flux_global().sum() == 0.0
    # escape everyone
    pe.step(np.array([5.0, 0.6, 0.7]), np.ones(1, np.int8), np.ones(1))
    f1 = pe.flux_global().sum()
    assert f1 > 0
    # step an escaped particle again (continue): walks from clipped pos
    pe.step(np.array([0.2, 0.6, 0.7]), np.ones(1, np.int8), np.ones(1))
    assert pe.flux_global().sum() > f1
    # coupled-host with zero flying residents
    gids = pe.resident_list()
    pe.step_local(np.zeros((len(gids), 3)).ravel(),
                  np.zeros(len(gids), np.int8), np.ones(len(gids)))
    # n_local mismatch must throw
    pe.resident_list()
    with pytest.raises(RuntimeError, match="resident_list"):
        pe.step_local(np.zeros(6), np.zeros(2, np.int8), np.ones(2))
    # step_local without a snapshot must throw
    pe2 = pt._core.PartitionedEngine(mesh, 1, device="cpu")
    pe2.localize(np.array([0.5, 0.5, 0.25]))
    pe2.step(np.array([0.5, 0.5, 0.3]), np.ones(1, np.int8), np.ones(1))
    with pytest.raises(RuntimeError, match="resident_list"):
        pe2.step_local(np.zeros(3), np.ones(1, np.int8), np.ones(1))


def test_periodic_world1_supported_and_worldN_documented(tmp_path):
    """Stateful engine + periodic: world-1 works (all pairs local);
    world>1 is a documented throw pointing at the stateless driver."""
    m = pt.build_box(3, 3, 3)
    fid, cen, nor = m.boundary_faces()
    hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
    m.set_periodic_faces(hi, lo, np.array([-1.0, 0.0, 0.0]))

    n = 40
    rng = np.random.default_rng(83)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = o.copy()
    d[:, 0] += rng.uniform(0.3, 1.2, n)  # wraps
    d[:, 1:] = np.clip(d[:, 1:], 0.05, 0.95)
    w = rng.uniform(0.5, 1.5, n)
    pe = pt._core.PartitionedEngine(m, n, device="cpu")
    pe.localize(o.ravel())
    pe.step(d.ravel(), np.ones(n, np.int8), w)
    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(pe.flux_global(), ref.flux(), atol=1e-12)

    # world>1 rejection: spawn two ranks and expect the documented error
    script = tmp_path / "w.py"
    script.write_text(r"""
import os
import numpy as np
import pumiumtally_amd as pt
m = pt.build_box(3, 3, 3)
fid, cen, nor = m.boundary_faces()
hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
m.set_periodic_faces(hi, lo, np.array([-1.0, 0.0, 0.0]))
try:
    pt._core.PartitionedEngine(m, 10, device="cpu")
except RuntimeError as e:
    assert "PartitionedTally" in str(e), e
    print("PERIODIC_REJECT_OK")
""")
    env = dict(os.environ)
    env.update({
        "WORLD_SIZE": "2", "MASTER_ADDR": "127.0.0.1",
        "PUMITALLY_PORT": str(27000 + (os.getpid() + 3) % 12000),
        "PUMITALLY_NO_TORCH": "1", "PYTHONPATH": ROOT,
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=120)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert "PERIODIC_REJECT_OK" in out, f"rank {r}:\n{out}"
