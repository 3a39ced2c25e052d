"""World-1 executions of every nccl(RCCL)-only code path on real hardware.

Round 1 shipped nccl branches (parallel/dist.py, parallel/partition.py)
that had only ever run shape-alike on gloo/CPU; any typo in them would
surface first on the driver's 8-GPU box.  These tests run each branch
live on ONE GPU: torch.distributed initialized with backend=nccl at
world_size 1 executes the same RCCL library calls (all_reduce,
all_gather_into_tensor, all_to_all_single) the 8-GPU run makes.
"""
import os
import subprocess
import sys

import numpy as np
import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os
import numpy as np
import torch
import torch.distributed as dist

import pumiumtally_amd as pt
from pumiumtally_amd.parallel import DistributedTally
from pumiumtally_amd.parallel.partition import PartitionedTally

assert torch.cuda.is_available()
mesh = pt.build_box(6, 6, 6)
n = 500
rng = np.random.default_rng(3)
o = rng.uniform(0.05, 0.95, size=(n, 3))
d = rng.uniform(0.05, 0.95, size=(n, 3))
w = rng.uniform(0.1, 1.0, n)

# oracle (CPU, single engine)
ref = pt.TallyEngine(mesh, n, device="cpu")
ref.copy_initial_position(o.ravel())
ref.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
want = ref.flux()

# --- DistributedTally over nccl (world 1): allreduce_flux + write path ---
dt = DistributedTally(mesh, n, backend="nccl")
assert dist.get_backend() == "nccl"
dt.copy_initial_position(o.ravel())
dt.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
got = dt.allreduce_flux()
assert np.allclose(got, want, atol=1e-9), np.abs(got - want).max()
print("NCCL_DIST_OK")

# --- PartitionedTally over nccl (world 1): device-resident rounds ---
ptal = PartitionedTally(mesh)
assert ptal._use_device_rounds()
ptal.run_segments(o, d, w)
gotp = ptal.flux_global()
assert np.allclose(gotp, want, atol=1e-9), np.abs(gotp - want).max()
print("NCCL_PART_OK")

# grouped + scored variants through the same device-resident path
ptal2 = PartitionedTally(mesh, ngroups=2, nscores=2)
g = rng.integers(0, 2, n).astype(np.uint16)
r2 = rng.uniform(0.5, 2.0, size=(n, 2))
ptal2.run_segments(o, d, w, groups=g, responses=r2)
ref2 = pt.TallyEngine(mesh, n, device="cpu", ngroups=2, nscores=2)
ref2.copy_initial_position(o.ravel())
ref2.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g,
          responses=r2)
assert np.allclose(ptal2.flux_global(), ref2.flux(), atol=1e-9)
print("NCCL_PART_SCORED_OK")

# --- the exact all_to_all_single packing partition.py uses, self-counts --
# (world 1: everything this rank sends comes back; a packing/offset bug
# corrupts the round-trip)
dev = torch.device("cuda:0")
rec_w = 9
k = 7
rec = torch.arange(k * rec_w, dtype=torch.float64, device=dev).view(k, rec_w)
counts = torch.tensor([k], dtype=torch.int64, device=dev)
world = dist.get_world_size()
all_counts = torch.zeros(world * world, dtype=torch.int64, device=dev)
dist.all_gather_into_tensor(all_counts, counts)
all_counts = all_counts.view(world, world)
in_counts = [int(c) * rec_w for c in all_counts[:, 0]]
out_counts = [int(c) * rec_w for c in counts]
recv = torch.empty(sum(in_counts), dtype=torch.float64, device=dev)
dist.all_to_all_single(recv, rec.view(-1), in_counts, out_counts)
assert torch.equal(recv.view(k, rec_w), rec)
print("NCCL_A2A_OK")

dist.destroy_process_group()
"""


@pytest.mark.gpu
def test_nccl_world1_branches(tmp_path):
    script = tmp_path / "worker.py"
    script.write_text(WORKER)
    env = dict(os.environ)
    env.update({
        "RANK": "0",
        "LOCAL_RANK": "0",
        "WORLD_SIZE": "1",
        "MASTER_ADDR": "127.0.0.1",
        "MASTER_PORT": str(24000 + (os.getpid() + 3) % 20000),
        "PYTHONPATH": ROOT,
    })
    out = subprocess.run([sys.executable, str(script)], env=env,
                         capture_output=True, text=True, timeout=600)
    assert out.returncode == 0, out.stdout + out.stderr
    for tag in ("NCCL_DIST_OK", "NCCL_PART_OK", "NCCL_PART_SCORED_OK",
                "NCCL_A2A_OK"):
        assert tag in out.stdout
