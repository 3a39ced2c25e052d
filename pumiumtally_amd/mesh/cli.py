"""Mesh CLI: the role of the Omega_h command-line utilities the reference
workflow depends on (msh2osh / osh_describe / osh_scale,
reference README.md:115-126).

    python -m pumiumtally_amd.mesh.cli convert  in.msh  out.osh
    python -m pumiumtally_amd.mesh.cli convert  in.msh  out.vtu   # viz export
    python -m pumiumtally_amd.mesh.cli describe mesh.osh
    python -m pumiumtally_amd.mesh.cli scale    in.osh out.osh --factor 0.01
    python -m pumiumtally_amd.mesh.cli box      out.osh --cells 55 --extent 1.0
    python -m pumiumtally_amd.mesh.cli partition mesh.osh --parts 8 \
        [--weights w.npy] [--out owners.vtu]

convert/scale/box write the format the output extension names: .osh
(directory, engine input), or .vtk/.vtu (viewer export with a "volume"
cell field).
"""
import argparse

import numpy as np


def _write_any(m, path):
    if path.endswith(".vtk") or path.endswith(".vtu"):
        m.write_vtk_fields(path, [("volume", np.asarray(m.volumes))])
    else:
        m.write_osh(path)


def main():
    ap = argparse.ArgumentParser(prog="pumiumtally_amd.mesh.cli")
    sub = ap.add_subparsers(dest="cmd", required=True)

    c = sub.add_parser("convert", help="Gmsh .msh -> .osh")
    c.add_argument("msh")
    c.add_argument("osh")

    d = sub.add_parser("describe", help="print mesh statistics")
    d.add_argument("mesh")

    s = sub.add_parser("scale", help="scale coordinates by a factor")
    s.add_argument("inp")
    s.add_argument("out")
    s.add_argument("--factor", type=float, required=True)

    b = sub.add_parser("box", help="generate a box mesh")
    b.add_argument("osh")
    b.add_argument("--cells", type=int, default=10)
    b.add_argument("--extent", type=float, default=1.0)

    p = sub.add_parser("partition",
                       help="preview a domain decomposition (Morton split)")
    p.add_argument("mesh")
    p.add_argument("--parts", type=int, default=8)
    p.add_argument("--ghost-rings", type=int, default=1)
    p.add_argument("--weights", default=None,
                   help=".npy per-element work estimates (weighted split)")
    p.add_argument("--out", default=None,
                   help="write owners as .npy / a .vtk|.vtu with an "
                        "'owner' cell field")

    args = ap.parse_args()
    import pumiumtally_amd as pt

    if args.cmd == "convert":
        m = pt.read_gmsh(args.msh)
        _write_any(m, args.osh)
        print(f"wrote {args.osh}: {m.nelems} tets, {m.nverts} verts")
    elif args.cmd == "describe":
        m = pt.read_mesh(args.mesh)
        v = m.volumes
        nbr = m.neighbors
        nb = int((nbr == -1).sum())
        print(f"elements : {m.nelems}")
        print(f"vertices : {m.nverts}")
        print(f"volume   : total {v.sum():.6g}, min {v.min():.3g}, max {v.max():.3g}")
        print(f"boundary : {nb} faces")
        c = m.coords
        print(f"bbox     : {c.min(axis=0)} .. {c.max(axis=0)}")
    elif args.cmd == "scale":
        m = pt.read_mesh(args.inp)
        coords = np.asarray(m.coords) * args.factor
        m2 = pt.mesh_from_arrays(coords, m.tet2vert)
        _write_any(m2, args.out)
        print(f"wrote {args.out} scaled by {args.factor}")
    elif args.cmd == "box":
        n = args.cells
        m = pt.build_box(n, n, n, args.extent, args.extent, args.extent)
        _write_any(m, args.osh)
        print(f"wrote {args.osh}: {m.nelems} tets")
    elif args.cmd == "partition":
        from pumiumtally_amd import _core

        m = pt.read_mesh(args.mesh)
        w = np.load(args.weights) if args.weights else None
        owners = _core.partition_morton(m, args.parts, w)
        counts = np.bincount(owners, minlength=args.parts)
        print(f"elements : {m.nelems} into {args.parts} parts")
        print(f"counts   : min {counts.min()}, max {counts.max()}, "
              f"imbalance {counts.max() / max(1, counts.min()):.3f}")
        if w is not None:
            sums = np.array([w[owners == r].sum() for r in range(args.parts)])
            print(f"work     : min {sums.min():.4g}, max {sums.max():.4g}, "
                  f"imbalance {sums.max() / max(1e-300, sums.min()):.3f}")
        ncut = nghost = 0
        for r in range(args.parts):
            sub = _core.extract_submesh(m, owners, r, args.ghost_rings)
            ncut += len(sub.foreign_gid)
            nghost += sub.local.nelems - int(counts[r])
        print(f"cut      : {ncut} handoff faces, {nghost} ghost elements "
              f"(rings={args.ghost_rings})")
        if args.out:
            if args.out.endswith(".npy"):
                np.save(args.out, owners)
            else:
                m.write_vtk_fields(args.out,
                                   [("owner", owners.astype(np.float64))])
            print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
