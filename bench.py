#!/usr/bin/env python3
"""Flagship benchmark: track-length tally throughput (particle-steps/sec).

BASELINE.json metric: particle-steps/sec (whole node) on a 1M-tet mesh.
One "step" = one MoveToNextLocation over the full particle batch: H2D
staging of origin/dest/flying/weights + the fused walk+tally kernel.
Synthetic straight-line histories (no network for datasets), fp64
positions/tallies (the reference's compute dtype).

Single GPU:   python bench.py --steps 20 --warmup 5
Multi-GPU:    python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
                  --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W
Scaling mode is WEAK: each rank owns its own --particles batch on a
replicated mesh; the only collective is one flux all-reduce at the end
(outside the timed region, like the reference which reduces only at
WriteTallyResults).
"""
import argparse
import json
import os
import sys
import time


class CommShim:
    """One surface for the two comm stacks: torch.distributed (RCCL via
    torch) or the library's own csrc/comm layer (RCCL via rcclComm, TCP
    fallback on CPU) -- bench runs identically on either."""

    def __init__(self, native: bool, world: int, local: int, on_gpu: bool):
        self.native = native
        self.world = world
        self.local = local
        self.on_gpu = on_gpu
        self._comm = None
        if native and world > 1:
            import pumiumtally_amd as pt
            self._comm = pt._core.make_native_comm(want_gpu=on_gpu,
                                                   device=local)

    def barrier(self):
        if self.world == 1:
            return
        if self.native:
            self._comm.barrier()
        else:
            import torch.distributed as dist
            dist.barrier()

    def max_scalar(self, v: float) -> float:
        if self.world == 1:
            return v
        if self.native:
            import numpy as np
            a = np.array([v])
            self._comm.allreduce_max(a)
            return float(a[0])
        import torch
        import torch.distributed as dist
        t = torch.tensor([v], dtype=torch.float64)
        if dist.get_backend() == "nccl":
            t = t.cuda(self.local)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        return float(t.cpu().item())

    def sum_array(self, a):
        if self.world == 1:
            return a
        if self.native:
            import numpy as np
            out = np.ascontiguousarray(a, dtype=np.float64)
            self._comm.allreduce_sum(out)
            return out
        import torch
        import torch.distributed as dist
        t = torch.from_numpy(a)
        if dist.get_backend() == "nccl":
            t = t.cuda(self.local)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        return t.cpu().numpy()

    def finalize(self):
        if self.world > 1 and not self.native:
            import torch.distributed as dist
            dist.destroy_process_group()
        self._comm = None


def _alloc(pt, shape, dtype, pin):
    """Pinned host array when the engine is on a GPU (pageable sources
    demote the chunked hipMemcpyAsync pipeline), plain numpy otherwise."""
    import numpy as np
    if pin:
        return pt.pinned_array(shape, dtype)
    return np.empty(shape, dtype)


def run_partitioned_stateful(args, mesh, cells, rank, world, local, device):
    """BASELINE config 3 (default partitioned mode): Morton element
    partition across ranks, ghost rings, cross-rank handoff over RCCL,
    with PERSISTENT particle residency (the native C++ PartitionedEngine,
    csrc/hip/partition_engine.hip).  One timed step = one step() over the
    global batch: upload dest/flying/weights, walk resident particles,
    exchange cut-crossers, repeat until converged.  All communication is
    the library's own comm layer (no torch in the data path)."""
    import numpy as np

    import pumiumtally_amd as pt
    from pumiumtally_amd.utils import make_box_histories

    import pumiumtally_amd as _pt
    n_global = args.particles * world
    pin = _pt.have_gpu() and device != "cpu"
    p0, p1, flying, weights = make_box_histories(
        (1.0, 1.0, 1.0), n_global, args.mean_chord, cells,
        seed=args.seed, pinned=pin, sort=not args.no_sort)
    # keep the pinned buffers as-is (reshape is a view, asarray/copy would
    # silently demote them to pageable and halve the step's H2D rate)
    o = np.asarray(p0).reshape(-1)
    d = np.asarray(p1).reshape(-1)
    w = np.asarray(weights)
    fly = np.asarray(flying)
    # groups/responses ride the same chunked H2D pipeline as dest/weights,
    # so they must be pinned too (pageable numpy arrays demote
    # hipMemcpyAsync to staged synchronous copies and serialize the step)
    groups = None
    if args.ngroups > 1:
        rng_g = np.random.default_rng(1234)
        groups = _alloc(pt, (n_global,), "uint16", pin)
        groups[:] = rng_g.integers(0, args.ngroups, n_global)

    responses = None
    if args.nscores > 1:
        rng_r = np.random.default_rng(4321)
        responses = _alloc(pt, (n_global, args.nscores), "float64", pin)
        responses[:] = rng_r.uniform(0.5, 2.0, size=(n_global, args.nscores))
    pe = pt._core.PartitionedEngine(mesh, n_global, device=device,
                                    ngroups=args.ngroups,
                                    nscores=args.nscores)
    pe.localize(o)
    ends = (o, d)

    def step(k):
        # ping-pong continue-mode: origin == committed position
        pe.step(ends[(k + 1) % 2], fly, w, groups=groups,
                responses=responses)

    def barrier_sync():
        pe.synchronize()
        pe.barrier()

    for k in range(args.warmup):
        step(k)
    barrier_sync()
    t_start = time.time()
    for k in range(args.warmup, args.warmup + args.steps):
        step(k)
    pe.synchronize()
    elapsed_local = time.time() - t_start
    barrier_sync()
    elapsed = float(pe.allreduce_max(
        np.array([elapsed_local], dtype=np.float64))[0])

    global_flux = pe.flux_global()
    st = pe.stats()
    if args.write_vtk and rank == 0:
        f = np.asarray(global_flux)
        if args.ngroups > 1:
            f = f.reshape(args.ngroups, mesh.nelems).sum(axis=0)
        pt.write_tally_vtk(args.write_vtk, mesh, f)
    if rank == 0:
        result = {
            "metric": "particle-steps/sec-partitioned",
            "value": n_global * args.steps / elapsed,
            "unit": "particle-steps/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic",
            "config": {
                "model": "track-length-tally-walk",
                "mesh_tets": int(mesh.nelems),
                "global_batch": n_global,
                "particles_per_gpu": args.particles,
                "mean_chord_elems": args.mean_chord,
                "seq_len": None,
                "parallelism": f"partitioned{world}-morton-ghost1-stateful",
                # steady-state steps use continue semantics (origin=None:
                # no particle resampled, 33 B/particle uploaded); the
                # replicated headline metric pays the reference-API origin
                # leg (57 B) -- metric names differ accordingly
                "step_mode": "continue",
                "comm": "native-" + pe.comm_kind,
                "ngroups": args.ngroups,
                "nscores": args.nscores,
                "device": "gpu" if device != "cpu" else "cpu",
                "resident_rank0": int(pe.resident),
                "lost_particles": st["lost_particles"],
                "flux_sum": float(np.asarray(global_flux).sum()),
            },
        }
        print(json.dumps(result), flush=True)


def run_partitioned(args, mesh, cells, rank, world, local, device):
    """Stateless legacy partitioned mode (--partitioned-stateless): the
    round-1 torch-based driver; every step re-localizes and re-uploads
    the whole global batch (kept for comparison; the stateful native
    engine above is the default --partitioned path)."""
    import numpy as np

    from pumiumtally_amd.parallel.partition import PartitionedTally
    from pumiumtally_amd.utils import make_box_histories

    n_global = args.particles * world
    p0, p1, flying, weights = make_box_histories(
        (1.0, 1.0, 1.0), n_global, args.mean_chord, cells,
        seed=args.seed, pinned=False, sort=not args.no_sort)
    o = np.asarray(p0).reshape(-1, 3)
    d = np.asarray(p1).reshape(-1, 3)
    w = np.asarray(weights)

    ptal = PartitionedTally(mesh, device=device, ngroups=args.ngroups)

    def barrier_sync():
        ptal.engine.synchronize()
        if world > 1:
            import torch.distributed as dist
            dist.barrier()

    for _ in range(args.warmup):
        ptal.run_segments(o, d, w)
    barrier_sync()
    t_start = time.time()
    for _ in range(args.steps):
        ptal.run_segments(o, d, w)
    ptal.engine.synchronize()
    elapsed_local = time.time() - t_start
    barrier_sync()
    if world > 1:
        import torch
        import torch.distributed as dist
        t = torch.tensor([elapsed_local], dtype=torch.float64)
        if dist.get_backend() == "nccl":
            t = t.cuda(local)
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.cpu().item())
    else:
        elapsed = elapsed_local

    global_flux = ptal.flux_global()
    if args.write_vtk:
        ptal.write_tally_results(args.write_vtk)
    if rank == 0:
        result = {
            "metric": "particle-steps/sec-partitioned",
            "value": n_global * args.steps / elapsed,
            "unit": "particle-steps/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic",
            "config": {
                "model": "track-length-tally-walk",
                "mesh_tets": int(mesh.nelems),
                "global_batch": n_global,
                "particles_per_gpu": args.particles,
                "mean_chord_elems": args.mean_chord,
                "seq_len": None,
                "parallelism": f"partitioned{world}-morton-ghost1",
                "ngroups": args.ngroups,
                "device": "gpu" if ptal.engine.is_gpu else "cpu",
                "lost_particles": ptal.engine.stats()["lost_particles"],
                "flux_sum": float(np.asarray(global_flux).sum()),
            },
        }
        print(json.dumps(result), flush=True)
    if world > 1:
        import torch.distributed as dist
        dist.destroy_process_group()


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1, help="informational; actual world size comes from torchrun env")
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--particles", type=int, default=10_000_000, help="particles per GPU")
    ap.add_argument("--mesh-tets", type=int, default=1_000_000)
    ap.add_argument("--mean-chord", type=float, default=8.0, help="target mean element crossings per step")
    ap.add_argument("--device", type=str, default=None, help="cpu / cuda:N (default: auto)")
    ap.add_argument("--seed", type=int, default=0)
    ap.add_argument("--write-vtk", type=str, default=None)
    ap.add_argument("--continue-mode", action="store_true",
                    help="use move_continue (no origin upload); NOT the headline "
                         "reference-API config -- reported with a distinct metric name")
    ap.add_argument("--no-sort", action="store_true", help="disable Morton ordering of particles")
    ap.add_argument("--source-frac", type=float, default=1.0,
                    help="origins sampled inside the central cube of this "
                         "axis fraction; small values (0.02) are the "
                         "BASELINE config-4 atomic-contention stress "
                         "(point source, hot elements); distinct metric "
                         "suffix when < 1")
    ap.add_argument("--ngroups", type=int, default=1,
                    help="energy groups (random per-particle group indices)")
    ap.add_argument("--nscores", type=int, default=1,
                    help="simultaneous tally scores (random per-particle "
                         "response multipliers); supported in replicated AND "
                         "partitioned modes")
    ap.add_argument("--backend", type=str, default=None,
                    help="torch.distributed backend override (nccl/gloo)")
    ap.add_argument("--device-resident", action="store_true",
                    help="inputs pre-staged in device memory (GPU transport-code "
                         "integration path); distinct metric name")
    ap.add_argument("--partitioned", action="store_true",
                    help="domain-decomposed mode (BASELINE config 3: Morton "
                         "element partition + ghost rings + cross-rank "
                         "particle handoff, persistent residency via the "
                         "native C++ PartitionedEngine); distinct metric name")
    ap.add_argument("--partitioned-stateless", action="store_true",
                    help="round-1 stateless partitioned driver (torch "
                         "collectives, per-step re-localization); kept for "
                         "comparison")
    ap.add_argument("--native-comm", action="store_true",
                    help="use the library's own comm layer (csrc/comm: "
                         "rcclComm over xGMI / TCP on CPU) instead of "
                         "torch.distributed; same metric, config notes the "
                         "stack")
    args = ap.parse_args()

    import numpy as np

    import pumiumtally_amd as pt
    from pumiumtally_amd.parallel import init_distributed
    from pumiumtally_amd.utils import make_box_histories

    if args.native_comm:
        if args.partitioned_stateless:
            raise SystemExit("--native-comm requires the stateful "
                             "partitioned mode (plain --partitioned)")
        rank = int(os.environ.get("RANK", "0"))
        world = int(os.environ.get("WORLD_SIZE", "1"))
        local = int(os.environ.get("LOCAL_RANK", rank))
    else:
        rank, world, local = init_distributed(args.backend)
    on_gpu = pt.have_gpu() and args.device != "cpu"
    if not on_gpu and args.device is None:
        # CPU fallback (debug only): shrink to something a serial walk finishes
        args.particles = min(args.particles, 20_000)
        args.mesh_tets = min(args.mesh_tets, 50_000)

    device = args.device or (f"cuda:{local}" if on_gpu else "cpu")

    t0 = time.time()
    from pumiumtally_amd.mesh import box_mesh_with_tets
    mesh, cells = box_mesh_with_tets(args.mesh_tets, extent=1.0)
    if rank == 0:
        print(f"[bench] mesh: {mesh.nelems} tets ({cells}^3 cells), built in {time.time()-t0:.1f}s",
              file=sys.stderr, flush=True)

    if args.partitioned_stateless:
        return run_partitioned(args, mesh, cells, rank, world, local, device)
    if args.partitioned:
        return run_partitioned_stateful(args, mesh, cells, rank, world, local,
                                        device)

    eng = pt.TallyEngine(mesh, args.particles, device=device,
                         ngroups=args.ngroups, nscores=args.nscores)
    p0, p1, flying, weights = make_box_histories(
        (1.0, 1.0, 1.0), args.particles, args.mean_chord, cells,
        seed=args.seed + rank, pinned=eng.is_gpu, sort=not args.no_sort,
        source_frac=args.source_frac)
    groups = None
    if args.ngroups > 1:
        rng_g = np.random.default_rng(1234 + rank)
        groups = _alloc(pt, (args.particles,), "uint16", eng.is_gpu)
        groups[:] = rng_g.integers(0, args.ngroups, args.particles)
    responses = None
    if args.nscores > 1:
        rng_r = np.random.default_rng(4321 + rank)
        responses = _alloc(pt, (args.particles, args.nscores), "float64",
                           eng.is_gpu)
        responses[:] = rng_r.uniform(0.5, 2.0,
                                     size=(args.particles, args.nscores))
    eng.copy_initial_position(p0.reshape(-1))
    eng.synchronize()

    ends = (p0.reshape(-1), p1.reshape(-1))

    if args.device_resident:
        import torch
        dev = torch.device(f"cuda:{local}")
        t_ends = (torch.from_numpy(np.asarray(ends[0])).to(dev),
                  torch.from_numpy(np.asarray(ends[1])).to(dev))
        t_flying = torch.from_numpy(np.asarray(flying)).to(dev)
        t_weights = torch.from_numpy(np.asarray(weights)).to(dev)
        torch.cuda.synchronize()  # materialize tensors once, up front

    def step(k):
        # ping-pong: walk P0->P1, then P1->P0; origin == current position so
        # phase A is a no-op compare, phase B walks the full segment set.
        if args.device_resident:
            eng.move_from_device(t_ends[(k + 1) % 2], t_flying, t_weights,
                                 sync_torch=False)
            return
        o, d = ends[k % 2], ends[(k + 1) % 2]
        if args.continue_mode:
            eng.move_continue(d, flying, weights, responses=responses)
        else:
            eng.move(o, d, flying, weights, groups=groups,
                     responses=responses)

    comm = CommShim(args.native_comm, world, local, on_gpu)

    def barrier_sync():
        eng.synchronize()
        comm.barrier()

    for k in range(args.warmup):
        step(k)
    barrier_sync()

    t_start = time.time()
    for k in range(args.warmup, args.warmup + args.steps):
        step(k)
    eng.synchronize()
    elapsed_local = time.time() - t_start
    barrier_sync()

    # MAX over ranks (slowest rank defines throughput)
    elapsed = comm.max_scalar(elapsed_local)

    stats = eng.stats()
    total_particle_steps = args.particles * args.steps * world
    value = total_particle_steps / elapsed

    # flux all-reduce + optional write (outside timed region, parity with
    # the reference's single reduction at WriteTallyResults)
    global_flux = comm.sum_array(eng.flux())
    if args.write_vtk and rank == 0:
        pt.write_tally_vtk(args.write_vtk, mesh, global_flux)

    if rank == 0:
        result = {
            "metric": "particle-steps/sec"
                      + ("-continue-mode" if args.continue_mode else "")
                      + ("-device-resident" if args.device_resident else "")
                      + ("-point-source" if args.source_frac < 1.0 else ""),
            "value": value,
            "unit": "particle-steps/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": elapsed / args.steps * 1000.0,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic",
            "config": {
                "model": "track-length-tally-walk",
                "mesh_tets": int(mesh.nelems),
                "global_batch": args.particles * world,
                "particles_per_gpu": args.particles,
                "mean_chord_elems": args.mean_chord,
                "seq_len": None,
                "parallelism": f"dp{world}-replicated-mesh",
                "source_frac": args.source_frac,
                "nscores": args.nscores,
                "comm": "native-rccl" if args.native_comm else "torch-rccl",
                "ngroups": args.ngroups,
                "device": "gpu" if eng.is_gpu else "cpu",
                "lost_particles": stats["lost_particles"],
                "flux_sum": float(global_flux.sum()),
            },
        }
        print(json.dumps(result), flush=True)
    comm.finalize()


if __name__ == "__main__":
    main()
