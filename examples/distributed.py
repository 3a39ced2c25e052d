"""Multi-GPU example: replicated-mesh data parallelism, stateless domain
decomposition, and the stateful native partitioned engine.

    python -m torch.distributed.run --nnodes=1 --nproc-per-node 8 \
        --master-addr 127.0.0.1 examples/distributed.py [--partitioned]

Stateful native engine (the library's OWN comm -- no torchrun needed,
plain processes with RANK/WORLD_SIZE env):

    for r in 0 1; do RANK=$r WORLD_SIZE=2 \
        python examples/distributed.py --stateful --device cpu & done; wait

CPU rehearsal (no GPUs, gloo):
    python -m torch.distributed.run --nnodes=1 --nproc-per-node 2 \
        --master-addr 127.0.0.1 examples/distributed.py --device cpu
"""
import argparse
import os
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import pumiumtally_amd as pt  # noqa: E402
from pumiumtally_amd.parallel import DistributedTally  # noqa: E402
from pumiumtally_amd.parallel.partition import PartitionedTally  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--partitioned", action="store_true",
                    help="stateless domain decomposition (torch collectives)")
    ap.add_argument("--stateful", action="store_true",
                    help="stateful native PartitionedEngine (library comm, "
                         "persistent particle residency)")
    ap.add_argument("--device", default=None)
    ap.add_argument("--particles", type=int, default=100_000)
    args = ap.parse_args()

    mesh = pt.build_box(12, 12, 12)
    rng = np.random.default_rng(0)

    if args.stateful:
        # Persistent residency: localize once, then step per transport
        # step; cut-crossers migrate over the library's own RCCL/TCP comm.
        n = args.particles
        pe = pt._core.PartitionedEngine(mesh, n,
                                        device=args.device or "auto")
        o = rng.uniform(0.02, 0.98, size=(n, 3))
        d = rng.uniform(0.02, 0.98, size=(n, 3))
        w = rng.uniform(0.2, 1.0, n)
        pe.localize(o.ravel())
        for step in range(3):
            dest = d if step % 2 == 0 else o
            pe.step(dest.ravel(), np.ones(n, np.int8), w)
        flux = pe.flux_global()
        if pe.rank == 0:
            print(f"[distributed] stateful world={pe.world} "
                  f"resident(rank0)={pe.resident} total flux {flux.sum():.4f}")
        return
    if args.partitioned:
        # True decomposition: every rank owns a Morton chunk of elements
        # (plus a ghost ring); all ranks pass the same global segment set
        # and ownership is resolved inside.
        tal = PartitionedTally(mesh, device=args.device)
        n = args.particles
        o = rng.uniform(0.02, 0.98, size=(n, 3))
        d = rng.uniform(0.02, 0.98, size=(n, 3))
        w = rng.uniform(0.2, 1.0, n)
        tal.run_segments(o, d, w)
        flux = tal.write_tally_results("flux_partitioned.vtk")
        rank = tal.rank
    else:
        # Replicated mesh, sharded particles: zero communication per step,
        # one all-reduce at the end.  The default (and faster) mode.
        tal = DistributedTally(mesh, args.particles, device=args.device)
        n = args.particles
        rng = np.random.default_rng(100 + tal.rank)  # each rank its own batch
        o = rng.uniform(0.02, 0.98, size=(n, 3))
        d = rng.uniform(0.02, 0.98, size=(n, 3))
        w = rng.uniform(0.2, 1.0, n)
        tal.copy_initial_position(o.ravel())
        tal.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
        flux = tal.write_tally_results("flux_replicated.vtk")
        rank = tal.rank

    if rank == 0:
        print(f"[distributed] world={tal.world} total flux {flux.sum():.4f}")
    if tal.world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
