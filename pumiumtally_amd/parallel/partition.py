"""Domain-decomposed tally: 8-way element partition + particle handoff.

For meshes that exceed what you want to replicate per GPU (the replicated
DistributedTally in .dist is the default and faster whenever the mesh
fits -- 288 GB HBM3E per MI355X fits multi-billion-tet walk data), this
mode partitions elements across ranks (Morton-balanced), builds per-rank
submeshes whose cut faces carry encoded foreign-element references, and
walks segments locally; particles crossing a cut are shipped to the
owning rank and resume mid-segment.  Shared-face planes are
bitwise-identical between submeshes (csrc/core/partition.cpp), so a
cross-rank walk tallies exactly what the single-mesh walk would.

Replaces the reference's pumipic picparts + ParticleTracer migration
(PumiTallyImpl.cpp:433-459,530-539) -- which is degenerate there (full
mesh on every rank, owners all rank 0) -- with a real decomposition.

Exchange collective: torch.distributed all_to_all_single over RCCL/xGMI
when on GPU (nccl backend), all_gather_object on gloo (CPU tests).

Exchange record (11+nscores float64 per handed-off particle when
responses are used, else 11): resume origin[3], destination[3], weight,
target global element id, energy group, walk progress t, previous
global element id[, response multipliers].  The (origin, t, prev)
triple is the bitwise handoff-resume state (csrc/core/walk.h
walk_segment doc): the receiving rank seeds its walk with the sender's
t-parametrization and replays the remaining crossings with identical
fp decisions, so partitioned flux attribution matches a single-mesh
walk elementwise.

Load balance: pass elem_weights to the constructor (per-element work
estimates -- e.g. a previous batch's raw flux) to split the Morton curve
by equal summed work instead of equal element count, or call
repartition(weights) between batches to rebuild the decomposition from
measured work (Morton chunks only balance element counts; weighted
splits balance particle work).
"""
from __future__ import annotations

import numpy as np

from .dist import init_distributed

_BASE_REC = 11  # floats per exchange record (before response columns)


class PartitionedTally:
    def __init__(self, mesh, device=None, backend=None, max_rounds: int = 64,
                 ghost_rings: int = 1, ngroups: int = 1, nscores: int = 1,
                 elem_weights=None):
        self.rank, self.world, self.local = init_distributed(backend)
        self.mesh = mesh
        self.max_rounds = max_rounds
        self.ngroups = max(1, int(ngroups))
        self.nscores = max(1, int(nscores))
        self.ghost_rings = ghost_rings
        self._device = device
        self._build(elem_weights)

    def _build(self, elem_weights=None):
        from .. import TallyEngine, _core, have_gpu

        mesh = self.mesh
        w = None if elem_weights is None else np.asarray(elem_weights, np.float64)
        self.owners = _core.partition_morton(mesh, self.world, w)
        self.sub = _core.extract_submesh(mesh, self.owners, self.rank,
                                         self.ghost_rings)
        self.l2g = self.sub.elem_l2g
        self.g2l = -np.ones(mesh.nelems, dtype=np.int64)
        self.g2l[self.l2g] = np.arange(len(self.l2g))
        self.foreign_gid = self.sub.foreign_gid
        self.foreign_owner = self.sub.foreign_owner
        # periodic cross-part faces carry a translation applied to BOTH the
        # shipped position and destination (zero rows for plain cuts)
        self.foreign_shift = np.asarray(self.sub.foreign_shift,
                                        np.float64).reshape(-1, 3)
        device = self._device
        if device is None:
            device = f"cuda:{self.local}" if have_gpu() else "cpu"
        # Engine over the local submesh; used only through walk_raw + flux.
        self.engine = TallyEngine(self.sub.local, 1, device=device,
                                  ngroups=self.ngroups,
                                  nscores=self.nscores)
        self._dev_tables = None  # lazy per-decomposition device lookup tables

    def repartition(self, elem_weights):
        """Rebuild the decomposition from per-element work estimates
        (global nelems array, identical on every rank -- e.g.
        flux_global().sum(axis=0) of the previous batch).  Discards the
        current local tally; call flux_global()/end-of-batch readout
        first."""
        elem_weights = np.asarray(elem_weights, np.float64).reshape(-1)
        if elem_weights.size != self.mesh.nelems:
            raise ValueError("elem_weights must have one entry per element")
        self._build(elem_weights)

    # -- helpers -----------------------------------------------------------
    def _exchange(self, records_per_rank, rec_w):
        """records_per_rank: list of world np.float64 arrays (k, rec_w);
        returns concatenated records received from all ranks."""
        if self.world == 1:
            return (records_per_rank[0] if records_per_rank
                    else np.zeros((0, rec_w)))
        import torch
        import torch.distributed as dist

        if dist.get_backend() == "nccl":
            send = torch.cat([torch.from_numpy(np.ascontiguousarray(r)).view(-1)
                              for r in records_per_rank]).cuda(self.local)
            in_counts = [r.size for r in records_per_rank]
            counts = torch.tensor(in_counts, dtype=torch.int64).cuda(self.local)
            all_counts = torch.zeros(self.world * self.world, dtype=torch.int64,
                                     device=counts.device)
            dist.all_gather_into_tensor(all_counts, counts)
            all_counts = all_counts.view(self.world, self.world).cpu().numpy()
            out_counts = list(all_counts[:, self.rank])
            recv = torch.empty(int(sum(out_counts)), dtype=torch.float64,
                               device=counts.device)
            dist.all_to_all_single(recv, send, out_counts, in_counts)
            return recv.cpu().numpy().reshape(-1, rec_w)
        # gloo: object all_gather
        gathered = [None] * self.world
        dist.all_gather_object(gathered, [np.asarray(r) for r in records_per_rank])
        mine = [g[self.rank] for g in gathered if g[self.rank].size]
        return (np.concatenate(mine).reshape(-1, rec_w) if mine
                else np.zeros((0, rec_w)))

    # -- public API --------------------------------------------------------
    def run_segments(self, origins, dests, weights, groups=None,
                     responses=None):
        """Walk one batch of global segments (origins->dests, weights),
        tallying into the partitioned flux.  Each rank passes the SAME
        global arrays (or its own shard -- ownership is resolved here);
        segments starting outside this rank's elements are ignored locally
        and handled by their owner.  groups: optional per-segment energy
        group indices (requires ngroups>1 at construction).  responses:
        optional (n, nscores) per-segment score multipliers (they ride the
        exchange record across ranks)."""
        origins = np.asarray(origins, np.float64).reshape(-1, 3)
        dests = np.asarray(dests, np.float64).reshape(-1, 3)
        weights = np.asarray(weights, np.float64).reshape(-1)
        if groups is not None:
            groups = np.asarray(groups, np.uint16).reshape(-1)
            if groups.size != weights.size:
                raise ValueError("groups size mismatch")
        if responses is not None:
            responses = np.asarray(responses, np.float64).reshape(
                -1, self.nscores)
            if responses.shape[0] != weights.size:
                raise ValueError("responses size mismatch")
        rec_w = _BASE_REC + (self.nscores if responses is not None else 0)
        gids = self.mesh.locate(origins)
        mine = (gids >= 0) & (self.owners[np.maximum(gids, 0)] == self.rank)
        pos = origins[mine]
        dst = dests[mine]
        wgt = weights[mine]
        grp = groups[mine] if groups is not None else None
        rsp = responses[mine] if responses is not None else None
        elem = self.g2l[gids[mine]].astype(np.int32)
        in_t = in_prev = None  # first round: fresh walks

        if self._use_device_rounds():
            return self._run_rounds_device(pos, dst, wgt, grp, rsp, elem,
                                           rec_w)

        l2g_arr = np.asarray(self.l2g, np.int64)
        for _round in range(self.max_rounds):
            outbound = [np.zeros((0, rec_w)) for _ in range(self.world)]
            if len(elem):
                (out_pos, out_elem, status, out_dest, out_o, out_t,
                 out_prev) = self.engine.walk_raw(
                    pos.ravel(), dst.ravel(), elem, wgt, grp, rsp,
                    in_t=in_t, in_prev=in_prev, resume=True)
                hand = status == 2
                if hand.any():
                    k = -(out_elem[hand].astype(np.int64) + 2)
                    tgt_gid = self.foreign_gid[k]
                    tgt_owner = self.foreign_owner[k]
                    shift = self.foreign_shift[k]
                    g_col = (grp[hand] if grp is not None
                             else np.zeros(int(hand.sum()), np.uint16))
                    ph = out_prev[hand].astype(np.int64)
                    prev_gid = np.where(ph >= 0, l2g_arr[np.maximum(ph, 0)],
                                        -1).astype(np.float64)
                    # resume state rides the record: wrap-segment origin
                    # (out_o) + progress t + exited-from element; out_dest,
                    # not dst, because in-walk reflective/periodic restarts
                    # mutate the destination
                    cols = [out_o[hand] + shift, out_dest[hand] + shift,
                            wgt[hand, None], tgt_gid[:, None].astype(np.float64),
                            g_col[:, None].astype(np.float64),
                            out_t[hand, None], prev_gid[:, None]]
                    if rsp is not None:
                        cols.append(rsp[hand])
                    rec = np.concatenate(cols, axis=1)
                    for r in range(self.world):
                        sel = tgt_owner == r
                        if sel.any():
                            outbound[r] = rec[sel]
            # global termination: exchange; empty everywhere -> done
            inbound = self._exchange(outbound, rec_w)
            if self.world > 1:
                import torch
                import torch.distributed as dist
                n_in = torch.tensor([float(inbound.size)])
                if dist.get_backend() == "nccl":
                    n_in = n_in.cuda(self.local)
                dist.all_reduce(n_in, op=dist.ReduceOp.SUM)
                if float(n_in.item()) == 0.0:
                    break
                total_pending = float(n_in.item())
            else:
                total_pending = inbound.size
            if total_pending == 0:
                break
            pos = inbound[:, 0:3]
            dst = inbound[:, 3:6]
            wgt = inbound[:, 6]
            elem = self.g2l[inbound[:, 7].astype(np.int64)].astype(np.int32)
            grp = (inbound[:, 8].astype(np.uint16)
                   if groups is not None else None)
            in_t = np.ascontiguousarray(inbound[:, 9])
            pg = inbound[:, 10].astype(np.int64)
            in_prev = np.where(pg >= 0, self.g2l[np.maximum(pg, 0)],
                               -1).astype(np.int32)
            rsp = (np.ascontiguousarray(inbound[:, 11:11 + self.nscores])
                   if responses is not None else None)
        else:
            raise RuntimeError("partitioned walk did not converge "
                               f"in {self.max_rounds} handoff rounds")

    def _use_device_rounds(self) -> bool:
        """Device-resident rounds: walk + record building + unpack stay in
        HBM; only per-round record counts touch the host.  Active on GPU
        engines when the exchange (if any) runs over RCCL (nccl) --
        gloo-coordinated GPU tests keep the host path.  PUMITALLY_PART_DEVICE=0
        disables."""
        import os
        if os.environ.get("PUMITALLY_PART_DEVICE", "1") == "0":
            return False
        if not self.engine.is_gpu:
            return False
        if self.world == 1:
            return True
        import torch.distributed as dist
        return dist.get_backend() == "nccl"

    def _run_rounds_device(self, pos, dst, wgt, grp, rsp, elem, rec_w):
        import torch

        dev = torch.device(f"cuda:{self.local}")
        to = lambda a: torch.from_numpy(np.ascontiguousarray(a)).to(dev)
        pos_t = to(pos)
        dst_t = to(dst)
        wgt_t = to(wgt)
        elem_t = to(elem)
        # uint16 group indices travel as bit-identical int16 tensors
        grp_t = to(grp.view(np.int16)) if grp is not None else None
        rsp_t = to(rsp) if rsp is not None else None
        if self._dev_tables is None:
            self._dev_tables = (
                to(self.g2l),
                to(np.asarray(self.foreign_gid, np.int64)),
                to(np.asarray(self.foreign_owner, np.int64)),
                to(self.foreign_shift),
                to(np.asarray(self.l2g, np.int64)),
            )
        g2l_t, fg_t, fo_t, fs_t, l2g_t = self._dev_tables
        in_t_t = in_prev_t = None  # first round: fresh walks

        for _round in range(self.max_rounds):
            k = int(elem_t.numel())
            if k:
                out_pos = torch.empty((k, 3), dtype=torch.float64, device=dev)
                out_elem = torch.empty(k, dtype=torch.int32, device=dev)
                status = torch.empty(k, dtype=torch.int8, device=dev)
                out_dest = torch.empty((k, 3), dtype=torch.float64,
                                       device=dev)
                out_o = torch.empty((k, 3), dtype=torch.float64, device=dev)
                out_t = torch.empty(k, dtype=torch.float64, device=dev)
                out_prev = torch.empty(k, dtype=torch.int32, device=dev)
                torch.cuda.synchronize()  # inputs/outputs materialized
                self.engine._eng.walk_raw_device(
                    k, pos_t.data_ptr(), dst_t.data_ptr(), elem_t.data_ptr(),
                    wgt_t.data_ptr(), out_pos.data_ptr(), out_elem.data_ptr(),
                    status.data_ptr(),
                    grp_t.data_ptr() if grp_t is not None else 0,
                    rsp_t.data_ptr() if rsp_t is not None else 0,
                    out_dest.data_ptr(),
                    in_t_t.data_ptr() if in_t_t is not None else 0,
                    in_prev_t.data_ptr() if in_prev_t is not None else 0,
                    out_o.data_ptr(), out_t.data_ptr(), out_prev.data_ptr())
                hand = status == 2
                nh = int(hand.sum())
            else:
                nh = 0
            if nh:
                idx = -(out_elem[hand].long() + 2)
                tgt_owner = fo_t[idx]
                shift = fs_t[idx]
                order = torch.argsort(tgt_owner)
                # out_dest, not dst: in-walk reflective/periodic restarts
                # mutate the destination
                ph = out_prev[hand].long()
                prev_gid = torch.where(ph >= 0, l2g_t[ph.clamp(min=0)],
                                       torch.full_like(ph, -1)).double()
                cols = [out_o[hand] + shift, out_dest[hand] + shift,
                        wgt_t[hand, None],
                        fg_t[idx][:, None].double(),
                        (grp_t[hand][:, None].double() if grp_t is not None
                         else torch.zeros((nh, 1), dtype=torch.float64,
                                          device=dev)),
                        out_t[hand, None], prev_gid[:, None]]
                if rsp_t is not None:
                    cols.append(rsp_t[hand])
                rec = torch.cat(cols, dim=1)[order].contiguous()
                counts = torch.bincount(tgt_owner, minlength=self.world)
            else:
                rec = torch.zeros((0, rec_w), dtype=torch.float64, device=dev)
                counts = torch.zeros(self.world, dtype=torch.int64,
                                     device=dev)

            if self.world == 1:
                inbound = rec
                if inbound.numel() == 0:
                    break
            else:
                import torch.distributed as dist
                all_counts = torch.zeros(self.world * self.world,
                                         dtype=torch.int64, device=dev)
                dist.all_gather_into_tensor(all_counts, counts)
                all_counts = all_counts.view(self.world, self.world)
                if int(all_counts.sum().item()) == 0:
                    break
                in_counts = [int(c) * rec_w for c in all_counts[:, self.rank]]
                out_counts = [int(c) * rec_w for c in counts]
                recv = torch.empty(sum(in_counts), dtype=torch.float64,
                                   device=dev)
                dist.all_to_all_single(recv, rec.view(-1), in_counts,
                                       out_counts)
                inbound = recv.view(-1, rec_w)
            pos_t = inbound[:, 0:3].contiguous()
            dst_t = inbound[:, 3:6].contiguous()
            wgt_t = inbound[:, 6].contiguous()
            elem_t = g2l_t[inbound[:, 7].long()].int().contiguous()
            if grp_t is not None:
                grp_t = inbound[:, 8].to(torch.int16).contiguous()
            in_t_t = inbound[:, 9].contiguous()
            pg = inbound[:, 10].long()
            in_prev_t = torch.where(pg >= 0, g2l_t[pg.clamp(min=0)],
                                    torch.full_like(pg, -1)).int().contiguous()
            if rsp_t is not None:
                rsp_t = inbound[:, 11:11 + self.nscores].contiguous()
        else:
            raise RuntimeError("partitioned walk did not converge "
                               f"in {self.max_rounds} handoff rounds")

    def flux_global(self) -> np.ndarray:
        """Scatter the local tally to global element ids and sum over ranks.
        Shape mirrors TallyEngine.flux(): (nelems,) plain, (ngroups, nelems)
        grouped, (nscores, nelems) scored, (nscores, ngroups, nelems) both."""
        local = np.asarray(self.engine.flux())
        ns, ng = self.nscores, self.ngroups
        nloc = len(self.l2g)
        out = np.zeros((ns, ng, self.mesh.nelems))
        out[:, :, self.l2g] = local.reshape(ns, ng, nloc)
        if self.world > 1:
            import torch
            import torch.distributed as dist
            t = torch.from_numpy(out)
            if dist.get_backend() == "nccl":
                t = t.cuda(self.local)
            dist.all_reduce(t, op=dist.ReduceOp.SUM)
            out = t.cpu().numpy()
        if ns == 1 and ng == 1:
            return out[0, 0]
        if ns == 1:
            return out[0]
        if ng == 1:
            return out[:, 0]
        return out

    def write_tally_pvtu(self, basename="fluxresult"):
        """Parallel VTK output: rank-owned .vtu pieces + a .pvtu master
        (the Omega_h vtk::write_parallel analog for a real
        decomposition).  Writes basename.pvtu + basename_p<r>.vtu;
        fields are the volume-normalized reduced flux (total + per-group
        / per-score when enabled)."""
        from .. import _core
        from ..mesh import write_pvtu

        f = np.asarray(self.flux_global()).reshape(
            self.nscores, self.ngroups, self.mesh.nelems)
        fields = []
        for k in range(self.nscores):
            name = "flux" if k == 0 else f"score{k}"
            fields.append((name, _core.normalize_flux(
                self.mesh, f[k].sum(axis=0))))
            if self.ngroups > 1:
                fields += [(f"{name}_g{g}", _core.normalize_flux(
                                self.mesh, f[k, g]))
                           for g in range(self.ngroups)]
        write_pvtu(basename, self.mesh, self.owners, self.rank,
                   self.world, fields)
        if self.world > 1:
            import torch.distributed as dist
            dist.barrier()

    def write_tally_results(self, filename="fluxresult.vtk"):
        from .. import _core, write_tally_vtk

        f = self.flux_global()
        if self.rank == 0:
            if self.nscores > 1:
                f2 = np.asarray(f).reshape(self.nscores, self.ngroups,
                                           self.mesh.nelems)
                fields = []
                for k in range(self.nscores):
                    name = "flux" if k == 0 else f"score{k}"
                    fields.append((name, _core.normalize_flux(
                        self.mesh, f2[k].sum(axis=0))))
                    if self.ngroups > 1:
                        fields += [(f"{name}_g{g}", _core.normalize_flux(
                                        self.mesh, f2[k, g]))
                                   for g in range(self.ngroups)]
                self.mesh.write_vtk_fields(filename, fields)
            elif self.ngroups > 1:
                fields = [("flux", _core.normalize_flux(self.mesh, f.sum(axis=0)))]
                fields += [(f"flux_g{g}", _core.normalize_flux(self.mesh, f[g]))
                           for g in range(self.ngroups)]
                self.mesh.write_vtk_fields(filename, fields)
            else:
                write_tally_vtk(filename, self.mesh, f)
        return f
