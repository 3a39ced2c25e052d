"""Domain-decomposition tests: submesh extraction and cross-part walks."""
import os
import subprocess
import sys

import numpy as np
import pytest

import pumiumtally_amd as pt
from pumiumtally_amd import _core


def test_partition_balanced():
    m = pt.build_box(4, 4, 4)
    owners = _core.partition_morton(m, 8)
    counts = np.bincount(owners, minlength=8)
    assert counts.sum() == m.nelems
    assert counts.max() - counts.min() <= 1


def test_submesh_structure():
    m = pt.build_box(3, 3, 3)
    owners = _core.partition_morton(m, 4)
    total = 0
    for p in range(4):
        sub = _core.extract_submesh(m, owners, p)
        total += sub.local.nelems
        assert np.array_equal(owners[sub.elem_l2g], np.full(sub.local.nelems, p))
        # local volumes match global volumes elementwise
        assert np.allclose(sub.local.volumes, m.volumes[sub.elem_l2g])
        # every foreign ref points to an element owned by another part
        assert (owners[sub.foreign_gid] != p).all()
        assert np.array_equal(owners[sub.foreign_gid], sub.foreign_owner)
    assert total == m.nelems


def test_single_rank_partitioned_equals_plain():
    """world=1 PartitionedTally (no exchange) == plain engine flux."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(3, 3, 3)
    n = 100
    rng = np.random.default_rng(5)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = rng.uniform(0.05, 0.95, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)

    ptal = PartitionedTally(m, device="cpu")
    ptal.run_segments(o, d, w)
    got = ptal.flux_global()

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(got, ref.flux(), atol=1e-12)


def test_walk_raw_handoff_encoding():
    """A walk on a submesh stops exactly at the partition cut."""
    m = pt.build_box(2, 1, 1, 2.0, 1.0, 1.0)
    # put x<1 cells in part 0, x>1 in part 1 (morton split does this for 2)
    owners = _core.partition_morton(m, 2)
    sub0 = _core.extract_submesh(m, owners, int(owners[m.locate(np.array([[0.2, 0.4, 0.5]]))[0]]))
    eng = pt.TallyEngine(sub0.local, 1, device="cpu")
    # segment crossing the whole domain
    start = np.array([[0.1, 0.4, 0.5]])
    lid = int(np.where(sub0.elem_l2g == m.locate(start)[0])[0][0])
    out_pos, out_elem, status, _ = eng._eng.walk_raw(
        start.ravel(), np.array([1.9, 0.4, 0.5]), np.array([lid], np.int32),
        np.ones(1))
    assert status[0] == 2
    k = -(int(out_elem[0]) + 2)
    assert 0 <= k < len(sub0.foreign_gid)
    # tally got only the inside part, and the handoff point is on the cut
    assert abs(eng.flux().sum() - (out_pos[0, 0] - 0.1)) < 1e-12


WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel.partition import PartitionedTally

rank = int(os.environ["RANK"])
mesh = pt.build_box(4, 4, 4)
n = 300
rng = np.random.default_rng(77)  # same segments on both ranks
o = rng.uniform(0.05, 0.95, size=(n, 3))
d = rng.uniform(0.05, 0.95, size=(n, 3))
w = rng.uniform(0.1, 1.0, n)

ptal = PartitionedTally(mesh, device="cpu", backend="gloo")
ptal.run_segments(o, d, w)
flux = ptal.flux_global()

if rank == 0:
    ref = pt.TallyEngine(mesh, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    err = np.abs(flux - ref.flux()).max()
    assert err < 1e-10, err
    print("PART_OK")
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_gloo_world2_partitioned(tmp_path):
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 9) % 20000),
        "WORLD_SIZE": "2",
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PART_OK" in outs[0]


@pytest.mark.gpu
def test_single_rank_partitioned_gpu():
    """walk_raw on GPU submesh == CPU, including handoff stops."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(6, 6, 6)
    n = 5000
    rng = np.random.default_rng(9)
    o = rng.uniform(0.02, 0.98, size=(n, 3))
    d = rng.uniform(0.02, 0.98, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)

    ptal = PartitionedTally(m, device="cuda:0")
    assert ptal.engine.is_gpu
    ptal.run_segments(o, d, w)
    got = ptal.flux_global()
    assert np.abs(got - ref.flux()).max() < 1e-10


def test_submesh_ghost_rings():
    m = pt.build_box(4, 4, 4)
    owners = _core.partition_morton(m, 4)
    for p in range(4):
        plain = _core.extract_submesh(m, owners, p)
        ghosted = _core.extract_submesh(m, owners, p, ghost_rings=1)
        n_owned = plain.local.nelems
        assert ghosted.local.nelems > n_owned
        # owned set is preserved
        assert set(plain.elem_l2g).issubset(set(ghosted.elem_l2g))
        # every ghost is a face-neighbor of the region grown so far and
        # owned elsewhere
        ghosts = set(ghosted.elem_l2g) - set(plain.elem_l2g)
        assert all(owners[g] != p for g in ghosts)
        # ghosting strictly reduces the cut surface per owned element
        # (foreign refs now sit one ring further out)
        assert (owners[ghosted.foreign_gid] != p).all()
        # volumes still match global
        assert np.allclose(ghosted.local.volumes, m.volumes[ghosted.elem_l2g])


WORKER_GPU = WORKER.replace('device="cpu", backend="gloo"',
                            'device="cuda:0", backend="gloo"')


@pytest.mark.gpu
def test_gloo_world2_partitioned_gpu(tmp_path):
    """Two ranks sharing one GPU (gloo rendezvous): the full partitioned
    walk/handoff path on device submeshes."""
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(WORKER_GPU)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 3) % 20000),
        "WORLD_SIZE": "2",
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = "0"
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=300)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PART_OK" in outs[0]


def test_weighted_partition_balance():
    """Work-weighted Morton split equalizes summed weight, not counts."""
    m = pt.build_box(6, 6, 6)
    rng = np.random.default_rng(3)
    # strongly skewed work: elements in x<0.5 cost 100x more
    cen = np.array([np.asarray(m.centroid(int(t))) for t in range(m.nelems)])
    w = np.where(cen[:, 0] < 0.5, 100.0, 1.0) * rng.uniform(0.5, 1.5, m.nelems)
    owners = _core.partition_morton(m, 4, w)
    assert owners.min() == 0 and owners.max() == 3
    sums = np.array([w[owners == p].sum() for p in range(4)])
    counts = np.bincount(owners, minlength=4)
    # weighted sums within 20% of each other; counts very unbalanced
    assert sums.max() / sums.min() < 1.2, sums
    assert counts.max() / counts.min() > 2, counts
    # unweighted call still balances counts
    owners0 = _core.partition_morton(m, 4)
    c0 = np.bincount(owners0, minlength=4)
    assert c0.max() - c0.min() <= 1


def test_single_rank_partitioned_groups():
    """world=1 PartitionedTally with energy groups == plain grouped engine."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(3, 3, 3)
    n = 200
    ng = 3
    rng = np.random.default_rng(11)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = rng.uniform(0.05, 0.95, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, ng, n).astype(np.uint16)

    ptal = PartitionedTally(m, device="cpu", ngroups=ng)
    ptal.run_segments(o, d, w, groups=g)
    got = ptal.flux_global()
    assert got.shape == (ng, m.nelems)

    ref = pt.TallyEngine(m, n, device="cpu", ngroups=ng)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g)
    assert np.allclose(got, ref.flux(), atol=1e-12)
    # group slices are genuinely distinct (skewed by construction)
    assert not np.allclose(got[0], got[1])


def test_repartition_weighted():
    """repartition(weights) rebuilds the decomposition and the new walk
    still matches the single-mesh oracle."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(3, 3, 3)
    n = 150
    rng = np.random.default_rng(21)
    o = rng.uniform(0.05, 0.95, size=(n, 3))
    d = rng.uniform(0.05, 0.95, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)

    ptal = PartitionedTally(m, device="cpu")
    ptal.run_segments(o, d, w)
    batch1 = ptal.flux_global()

    # feed measured work back in; decomposition changes, results don't
    ptal.repartition(batch1 + 1e-9)
    ptal.run_segments(o, d, w)
    batch2 = ptal.flux_global()

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(batch1, ref.flux(), atol=1e-12)
    assert np.allclose(batch2, ref.flux(), atol=1e-12)


WORKER_GROUPS = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel.partition import PartitionedTally

rank = int(os.environ["RANK"])
mesh = pt.build_box(4, 4, 4)
n = 300
ng = 2
rng = np.random.default_rng(78)  # same segments on both ranks
o = rng.uniform(0.05, 0.95, size=(n, 3))
d = rng.uniform(0.05, 0.95, size=(n, 3))
w = rng.uniform(0.1, 1.0, n)
g = rng.integers(0, ng, n).astype(np.uint16)

ptal = PartitionedTally(mesh, device="cpu", backend="gloo", ngroups=ng)
ptal.run_segments(o, d, w, groups=g)
flux = ptal.flux_global()

if rank == 0:
    ref = pt.TallyEngine(mesh, n, device="cpu", ngroups=ng)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g)
    err = np.abs(flux - ref.flux()).max()
    assert err < 1e-10, err
    print("PART_OK")
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_gloo_world2_partitioned_groups(tmp_path):
    """Cross-rank handoff carries the energy group (9-double record)."""
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(WORKER_GROUPS)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 57) % 20000),
        "WORLD_SIZE": "2",
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PART_OK" in outs[0]


@pytest.mark.gpu
def test_single_rank_partitioned_groups_gpu():
    """Grouped walk_raw on a GPU submesh == grouped CPU oracle."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(6, 6, 6)
    n = 5000
    ng = 4
    rng = np.random.default_rng(13)
    o = rng.uniform(0.02, 0.98, size=(n, 3))
    d = rng.uniform(0.02, 0.98, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, ng, n).astype(np.uint16)

    ref = pt.TallyEngine(m, n, device="cpu", ngroups=ng)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g)

    ptal = PartitionedTally(m, device="cuda:0", ngroups=ng)
    assert ptal.engine.is_gpu
    ptal.run_segments(o, d, w, groups=g)
    got = ptal.flux_global()
    assert np.abs(got - ref.flux()).max() < 1e-10


def test_submesh_carries_face_bc():
    """Per-face reflective bits survive submesh extraction."""
    m = pt.build_box(3, 3, 3)
    fid, cen, nor = m.boundary_faces()
    plus_x = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
    m.set_reflective_faces(plus_x)
    owners = _core.partition_morton(m, 2)
    marked = 0
    for p in range(2):
        sub = _core.extract_submesh(m, owners, p, ghost_rings=1)
        lfid, lcen, lnor = sub.local.boundary_faces()
        for i in range(len(lfid)):
            if abs(lcen[i, 0] - 1.0) < 1e-12:
                assert sub.local.face_is_reflective(int(lfid[i]))
                marked += 1
            else:
                assert not sub.local.face_is_reflective(int(lfid[i]))
    # ghost rings can replicate a boundary face in both submeshes
    assert marked >= len(plus_x)


@pytest.mark.gpu
def test_partitioned_device_rounds_gpu():
    """Device-resident round loop (walk_raw_device + on-device record
    building) == the host-staged loop == the single-mesh oracle."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(6, 6, 6)
    n, G, S = 8000, 2, 2
    rng = np.random.default_rng(61)
    o = rng.uniform(0.02, 0.98, size=(n, 3))
    d = rng.uniform(0.02, 0.98, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, G, n).astype(np.uint16)
    r = rng.uniform(0.2, 2.0, size=(n, S))

    ref = pt.TallyEngine(m, n, device="cpu", ngroups=G, nscores=S)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g,
             responses=r)

    ptal = PartitionedTally(m, device="cuda:0", ngroups=G, nscores=S)
    assert ptal.engine.is_gpu and ptal._use_device_rounds()
    ptal.run_segments(o, d, w, groups=g, responses=r)
    got = ptal.flux_global()
    assert np.abs(got - ref.flux()).max() < 1e-10

    # host-staged loop agrees (PUMITALLY_PART_DEVICE=0 route)
    import os
    os.environ["PUMITALLY_PART_DEVICE"] = "0"
    try:
        ptal2 = PartitionedTally(m, device="cuda:0", ngroups=G, nscores=S)
        assert not ptal2._use_device_rounds()
        ptal2.run_segments(o, d, w, groups=g, responses=r)
        assert np.abs(ptal2.flux_global() - ref.flux()).max() < 1e-10
    finally:
        del os.environ["PUMITALLY_PART_DEVICE"]


@pytest.mark.gpu
def test_partitioned_device_rounds_plain_gpu():
    """Plain (ungrouped, unscored) device rounds vs oracle."""
    from pumiumtally_amd.parallel.partition import PartitionedTally

    m = pt.build_box(5, 5, 5)
    n = 6000
    rng = np.random.default_rng(62)
    o = rng.uniform(0.02, 0.98, size=(n, 3))
    d = rng.uniform(0.02, 0.98, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)

    ptal = PartitionedTally(m, device="cuda:0")
    ptal.run_segments(o, d, w)
    assert np.abs(ptal.flux_global() - ref.flux()).max() < 1e-10


def test_weighted_partition_never_empty_under_skew():
    """One element holding almost all the weight must not starve other
    parts: the split clamp guarantees every part owns >= 1 element
    (ADVICE round-1, partition.cpp weighted split)."""
    m = pt.build_box(3, 3, 3)
    w = np.full(m.nelems, 1e-12)
    w[0] = 1e12  # extreme skew: element 0 dominates the total
    for nparts in (2, 4, 8):
        owners = np.asarray(pt._core.partition_morton(m, nparts, w))
        counts = np.bincount(owners, minlength=nparts)
        assert counts.min() >= 1, counts
    # more parts than elements is a clear error, not an empty submesh
    tiny = pt.build_box(1, 1, 1)  # 6 tets
    with pytest.raises(RuntimeError):
        pt._core.partition_morton(tiny, 7, None)


PERIODIC_WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel.partition import PartitionedTally

rank = int(os.environ["RANK"])
mesh = pt.build_box(3, 3, 3)
fid, cen, nor = mesh.boundary_faces()
hi = fid[np.abs(cen[:, 0] - 1.0) < 1e-12]
lo = fid[np.abs(cen[:, 0] - 0.0) < 1e-12]
mesh.set_periodic_faces(hi, lo, np.array([-1.0, 0.0, 0.0]))

n = 200
rng = np.random.default_rng(13)  # same segments on both ranks
o = rng.uniform(0.05, 0.95, size=(n, 3))
d = o.copy()
d[:, 0] += rng.uniform(0.2, 1.3, n)  # wrap through x=1
d[:, 1:] = np.clip(d[:, 1:] + rng.normal(0, 0.15, size=(n, 2)), 0.02, 0.98)
w = rng.uniform(0.1, 1.0, n)

ptal = PartitionedTally(mesh, device="cpu", backend="gloo")
ptal.run_segments(o, d, w)
flux = ptal.flux_global()

if rank == 0:
    ref = pt.TallyEngine(mesh, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    err = np.abs(flux - ref.flux()).max()
    assert err < 1e-10, err
    print("PART_PERIODIC_OK")
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_gloo_world2_partitioned_periodic(tmp_path):
    """Cross-part periodic faces through the full PartitionedTally
    driver (round-1 closed this out): the handoff record applies the
    pair translation to position and destination (foreign_shift)."""
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(PERIODIC_WORKER)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 171) % 20000),
        "WORLD_SIZE": "2",
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PART_PERIODIC_OK" in outs[0]


REFLECT_WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel.partition import PartitionedTally

os.environ["PUMITALLY_BC"] = "reflective"
rank = int(os.environ["RANK"])
mesh = pt.build_box(3, 3, 3)

n = 150
rng = np.random.default_rng(29)
o = rng.uniform(0.05, 0.95, size=(n, 3))
d = o + rng.normal(0, 0.7, size=(n, 3))  # many exits -> reflections

ptal = PartitionedTally(mesh, device="cpu", backend="gloo")
w = rng.uniform(0.1, 1.0, n)
ptal.run_segments(o, d, w)
flux = ptal.flux_global()

if rank == 0:
    ref = pt.TallyEngine(mesh, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    err = np.abs(flux - ref.flux()).max()
    assert err < 1e-10, err
    # conservation under reflection: full segment lengths tallied
    seg = np.linalg.norm(d - o, axis=1)
    assert abs(flux.sum() - (seg * w).sum()) < 1e-9
    print("PART_REFLECT_OK")
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_gloo_world2_partitioned_reflective(tmp_path):
    """Reflect-then-handoff: a reflective restart mutates the walk's
    destination; the exchange record must resume toward the MUTATED
    destination (walk_raw out_dest), or cross-cut reflections tally the
    wrong remainder (same bug class as the periodic ping-pong)."""
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(REFLECT_WORKER)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 217) % 20000),
        "WORLD_SIZE": "2",
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PART_REFLECT_OK" in outs[0]


def test_pvtu_pieces_tile_mesh(tmp_path):
    """write_tally_pvtu: rank-owned pieces + master; pieces' cell counts
    sum to nelems, piece volumes sum to the box volume, and the master
    references every piece."""
    from pumiumtally_amd.mesh import write_pvtu
    from pumiumtally_amd import _core

    m = pt.build_box(3, 3, 3)
    owners = np.asarray(_core.partition_morton(m, 2))
    flux = np.arange(m.nelems, dtype=np.float64)
    base = str(tmp_path / "out")
    for r in range(2):  # emulate both ranks in-process
        write_pvtu(base, m, owners, r, 2, [("flux", flux)])
    master = (tmp_path / "out.pvtu").read_text()
    assert 'Piece Source="out_p0.vtu"' in master
    assert 'Piece Source="out_p1.vtu"' in master
    # structural check of the pieces: parse NumberOfCells from each
    total = 0
    for r in range(2):
        txt = (tmp_path / f"out_p{r}.vtu").read_bytes().decode("latin1")
        import re
        mobj = re.search(r'NumberOfCells="(\d+)"', txt)
        total += int(mobj.group(1))
        assert f'Name="flux"' in txt
    assert total == m.nelems


def test_per_face_reflective_carried_into_submesh():
    """Per-face reflective BCs survive extract_submesh (the boundary-
    condition bit is carried onto the local face), so a partitioned walk
    reflects exactly like the full-mesh walk."""
    m = pt.build_box(3, 3, 3)
    fid, cen, nor = m.boundary_faces()
    xlo = fid[np.abs(cen[:, 0]) < 1e-12]
    m.set_reflective_faces(xlo)  # mirror only the x=0 wall

    n = 80
    rng = np.random.default_rng(19)
    o = rng.uniform(0.05, 0.4, size=(n, 3))
    d = o.copy()
    d[:, 0] -= rng.uniform(0.2, 0.8, n)  # drive into the mirrored wall
    w = rng.uniform(0.5, 1.5, n)

    ref = pt.TallyEngine(m, n, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)

    from pumiumtally_amd.parallel.partition import PartitionedTally
    ptal = PartitionedTally(m, device="cpu")
    ptal.run_segments(o, d, w)
    got = ptal.flux_global()
    assert np.allclose(got, ref.flux(), atol=1e-12), \
        np.abs(got - ref.flux()).max()
    # the mirrored wall conserves those segments entirely
    seg = np.linalg.norm(d - o, axis=1)
    assert abs(got.sum() - (seg * w).sum()) < 1e-9

    # and through the stateful engine
    pe = pt._core.PartitionedEngine(m, n, device="cpu")
    pe.localize(o.ravel())
    pe.step(d.ravel(), np.ones(n, np.int8), w)
    assert np.allclose(pe.flux_global(), ref.flux(), atol=1e-12)


PVTU_WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel.partition import PartitionedTally

rank = int(os.environ["RANK"])
mesh = pt.build_box(3, 3, 3)
n = 100
rng = np.random.default_rng(23)
o = rng.uniform(0.05, 0.95, size=(n, 3))
d = rng.uniform(0.05, 0.95, size=(n, 3))
w = rng.uniform(0.1, 1.0, n)
ptal = PartitionedTally(mesh, device="cpu", backend="gloo")
ptal.run_segments(o, d, w)
ptal.write_tally_pvtu(os.environ["PT_BASE"])
if rank == 0:
    print("PVTU_OK")
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_pvtu_world2(tmp_path):
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(PVTU_WORKER)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 311) % 20000),
        "WORLD_SIZE": "2",
        "PT_BASE": str(tmp_path / "flux"),
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE,
                                      stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=180)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "PVTU_OK" in outs[0]
    assert (tmp_path / "flux.pvtu").exists()
    assert (tmp_path / "flux_p0.vtu").exists()
    assert (tmp_path / "flux_p1.vtu").exists()
    import re
    total = 0
    for r in range(2):
        txt = (tmp_path / f"flux_p{r}.vtu").read_bytes().decode("latin1")
        total += int(re.search(r'NumberOfCells="(\d+)"', txt).group(1))
    m = pt.build_box(3, 3, 3)
    assert total == m.nelems
