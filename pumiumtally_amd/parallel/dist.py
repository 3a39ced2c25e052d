"""Multi-GPU scaling over RCCL (torch.distributed) / gloo.

MI355X-first scaling design (SURVEY.md sections 2.3/2.5): the reference's
domain decomposition is degenerate (every rank holds the full mesh, all
elements owned by rank 0).  On MI355X, 288 GB of HBM3E per GPU makes real
domain decomposition unnecessary for any mesh that fits a node: a 10M-tet
mesh is ~1.5 GB of walk data.  The native strategy is therefore

    * replicate the mesh on every GPU (one process per GPU),
    * data-parallel over particles: each rank owns its shard of the
      particle batch and walks it with ZERO communication per step,
    * one all-reduce (sum) of the nelems-sized flux tally before
      normalization/writing -- the only collective in the whole run.

This replaces the reference's pumipic picparts + per-step migration
machinery (PumiTallyImpl.cpp:111-145,433-459,530-539) with a design whose
communication volume is independent of both the particle count and the
step count.  An explicit 8-way element-partition mode with RCCL
all-to-all particle exchange lives in pumiumtally_amd.parallel.partition
for meshes that exceed a single GPU's memory.

Backend: "nccl" (RCCL over xGMI) when GPUs are present, "gloo" for
CPU-only runs and the CI world_size=2 tests.
"""
from __future__ import annotations

import os
from typing import Optional

import numpy as np


def init_distributed(backend: Optional[str] = None):
    """Initialize torch.distributed from torchrun env vars; returns
    (rank, world_size, local_rank).  Safe to call in single-process mode
    (returns (0, 1, 0) without initializing).  An EXPLICIT backend forces
    initialization even at world_size 1 (when RANK is set) so the
    nccl-only code paths can be executed live on a single GPU."""
    if "RANK" not in os.environ or (
            backend is None and int(os.environ.get("WORLD_SIZE", "1")) == 1):
        return 0, 1, 0  # single process: torch never imported
    import torch
    import torch.distributed as dist

    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    if not dist.is_initialized():
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        dist.init_process_group(backend=backend)
    rank = dist.get_rank()
    world = dist.get_world_size()
    local = int(os.environ.get("LOCAL_RANK", rank))
    if backend == "nccl":
        torch.cuda.set_device(local)
    return rank, world, local


class DistributedTally:
    """Replicated-mesh, particle-data-parallel tally engine.

    Each rank constructs its own TallyEngine on its GPU (or CPU) and walks
    its own particles.  `allreduce_flux()` sums the per-rank tallies; rank 0
    can then write the global VTK result.
    """

    def __init__(self, mesh, particles_per_rank: int, device: Optional[str] = None,
                 backend: Optional[str] = None, ngroups: int = 1,
                 nscores: int = 1):
        from .. import TallyEngine, have_gpu

        self.rank, self.world, self.local = init_distributed(backend)
        if device is None:
            device = f"cuda:{self.local}" if have_gpu() else "cpu"
        self.device = device
        self.engine = TallyEngine(mesh, particles_per_rank, device=device,
                                  ngroups=ngroups, nscores=nscores)
        self.mesh = mesh

    def copy_initial_position(self, positions):
        self.engine.copy_initial_position(positions)

    def move(self, origin, dest, flying, weights, groups=None,
             responses=None):
        self.engine.move(origin, dest, flying, weights, groups=groups,
                         responses=responses)

    def barrier(self):
        if self.world > 1:
            import torch.distributed as dist

            dist.barrier()

    def allreduce_flux(self) -> np.ndarray:
        """Sum the flux tallies across ranks; returns the global tally."""
        self.engine.synchronize()
        local = self.engine.flux()
        if self.world == 1:
            return local
        import torch
        import torch.distributed as dist

        use_gpu = dist.get_backend() == "nccl"
        t = torch.from_numpy(local)
        if use_gpu:
            t = t.cuda(self.local)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return t.cpu().numpy()

    def write_tally_results(self, filename: str = "fluxresult.vtk"):
        """All-reduce, then rank 0 writes the normalized global tally
        (grouped tallies get per-group fields, via the engine's writer)."""
        global_flux = self.allreduce_flux()
        if self.rank == 0:
            local = self.engine.flux()
            self.engine.set_flux(global_flux)
            self.engine.write_tally_results(filename)
            self.engine.set_flux(local)  # restore this rank's partial tally
        self.barrier()
        return global_flux
