"""Distributed correctness over gloo (world_size=2, CPU).

The multi-GPU path (one process per GPU over RCCL) is identical code with
backend "nccl"; this pins its correctness by construction: sharded
particles + flux all-reduce == single-engine global tally, exactly.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

import pumiumtally_amd as pt

WORKER = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel import DistributedTally

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
mesh = pt.build_box(4, 4, 4)
n_total = 200
n_rank = n_total // world

rng = np.random.default_rng(123)  # same stream everywhere
o = rng.uniform(0.05, 0.95, size=(n_total, 3))
d = rng.uniform(0.05, 0.95, size=(n_total, 3))
w = rng.uniform(0.1, 1.0, n_total)

lo, hi = rank * n_rank, (rank + 1) * n_rank
dt = DistributedTally(mesh, n_rank, device="cpu", backend="gloo")
dt.copy_initial_position(o[lo:hi].ravel())
dt.move(o[lo:hi].ravel(), d[lo:hi].ravel(),
        np.ones(n_rank, np.int8), w[lo:hi])
global_flux = dt.allreduce_flux()

if rank == 0:
    ref = pt.TallyEngine(mesh, n_total, device="cpu")
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n_total, np.int8), w)
    assert np.allclose(global_flux, ref.flux(), atol=1e-12), \
        np.abs(global_flux - ref.flux()).max()
    out = dt.write_tally_results(os.environ["PT_TEST_OUT"])
    assert np.allclose(out, ref.flux(), atol=1e-12)
    print("DIST_OK")
else:
    dt.write_tally_results(os.environ["PT_TEST_OUT"])
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_gloo_world2_flux_allreduce(tmp_path):
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 7) % 20000),
        "WORLD_SIZE": "2",
        "PT_TEST_OUT": str(tmp_path / "flux.vtk"),
        "PYTHONPATH": os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen([sys.executable, str(script)], env=e,
                                      stdout=subprocess.PIPE, stderr=subprocess.STDOUT))
    outs = []
    for p in procs:
        out, _ = p.communicate(timeout=180)
        outs.append(out.decode())
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    assert "DIST_OK" in outs[0]
    assert (tmp_path / "flux.vtk").exists()


def test_bench_single_process_cpu(tmp_path):
    """bench.py driver contract: one JSON line with the required keys."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "bench.py"), "--steps", "2",
         "--warmup", "1", "--particles", "2000", "--mesh-tets", "3000",
         "--device", "cpu"],
        capture_output=True, text=True, timeout=300, cwd=root)
    assert out.returncode == 0, out.stderr
    line = out.stdout.strip().splitlines()[-1]
    r = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in r, key
    assert r["n_gpus"] == 1 and r["steps"] == 2
    assert r["value"] > 0
    assert r["config"]["lost_particles"] == 0


def test_bench_gloo_world2(tmp_path):
    """bench.py under torchrun-style env (world 2, gloo): the exact launch
    contract the driver's SCALE run uses, minus the GPUs."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 123) % 20000),
        "WORLD_SIZE": "2",
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(root, "bench.py"), "--steps", "2",
             "--warmup", "1", "--particles", "1500", "--mesh-tets", "3000",
             "--device", "cpu", "--backend", "gloo", "--gpus", "2"],
            env=e, cwd=root, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=300)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    # rank 0 prints the single JSON result line
    line = [l for l in outs[0].splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["n_gpus"] == 2
    assert res["config"]["global_batch"] == 3000
    # rank 1 prints nothing JSON
    assert not any(l.startswith("{") for l in outs[1].splitlines())


WORKER_SCORED = r"""
import os
import numpy as np
import pumiumtally_amd as pt
from pumiumtally_amd.parallel import DistributedTally

rank = int(os.environ["RANK"])
world = int(os.environ["WORLD_SIZE"])
mesh = pt.build_box(4, 4, 4)
n_total = 200
n_rank = n_total // world
G, S = 2, 2

rng = np.random.default_rng(321)  # same stream everywhere
o = rng.uniform(0.05, 0.95, size=(n_total, 3))
d = rng.uniform(0.05, 0.95, size=(n_total, 3))
w = rng.uniform(0.1, 1.0, n_total)
g = rng.integers(0, G, n_total).astype(np.uint16)
r = rng.uniform(0.2, 2.0, size=(n_total, S))

lo, hi = rank * n_rank, (rank + 1) * n_rank
dt = DistributedTally(mesh, n_rank, device="cpu", backend="gloo",
                      ngroups=G, nscores=S)
dt.copy_initial_position(o[lo:hi].ravel())
dt.move(o[lo:hi].ravel(), d[lo:hi].ravel(), np.ones(n_rank, np.int8),
        w[lo:hi], groups=g[lo:hi], responses=np.ascontiguousarray(r[lo:hi]))
global_flux = dt.allreduce_flux()

if rank == 0:
    ref = pt.TallyEngine(mesh, n_total, device="cpu", ngroups=G, nscores=S)
    ref.copy_initial_position(o.ravel())
    ref.move(o.ravel(), d.ravel(), np.ones(n_total, np.int8), w, groups=g,
             responses=r)
    assert global_flux.shape == (S, G, mesh.nelems)
    assert np.allclose(global_flux, ref.flux(), atol=1e-12), \
        np.abs(global_flux - ref.flux()).max()
    print("DIST_SCORED_OK")
import torch.distributed as dist
dist.destroy_process_group()
"""


def test_gloo_world2_scored_grouped(tmp_path):
    """Replicated driver with groups+scores: all-reduced (S,G,nelems)
    tally equals the single-engine oracle."""
    pytest.importorskip("torch")
    script = tmp_path / "worker.py"
    script.write_text(WORKER_SCORED)
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 201) % 20000),
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
    assert "DIST_SCORED_OK" in outs[0]


def test_bench_partitioned_gloo_world2(tmp_path):
    """bench.py --partitioned (BASELINE config 3 shape) under world-2 gloo."""
    import json
    import subprocess
    import sys

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.update({
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(20000 + (os.getpid() + 307) % 20000),
        "WORLD_SIZE": "2",
    })
    procs = []
    for r in range(2):
        e = dict(env)
        e["RANK"] = str(r)
        e["LOCAL_RANK"] = str(r)
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(root, "bench.py"), "--partitioned",
             "--steps", "2", "--warmup", "1", "--particles", "800",
             "--mesh-tets", "3000", "--device", "cpu", "--backend", "gloo"],
            env=e, cwd=root, stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT))
    outs = [p.communicate(timeout=300)[0].decode() for p in procs]
    for r, (p, out) in enumerate(zip(procs, outs)):
        assert p.returncode == 0, f"rank {r} failed:\n{out}"
    line = [l for l in outs[0].splitlines() if l.startswith("{")][-1]
    res = json.loads(line)
    assert res["metric"] == "particle-steps/sec-partitioned"
    assert res["n_gpus"] == 2 and res["config"]["global_batch"] == 1600
    assert res["config"]["lost_particles"] == 0
