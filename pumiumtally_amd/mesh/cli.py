"""Mesh CLI: the role of the Omega_h command-line utilities the reference
workflow depends on (msh2osh / osh_describe / osh_scale,
reference README.md:115-126).

    python -m pumiumtally_amd.mesh.cli convert  in.msh  out.osh
    python -m pumiumtally_amd.mesh.cli convert  in.msh  out.vtu   # viz export
    python -m pumiumtally_amd.mesh.cli describe mesh.osh
    python -m pumiumtally_amd.mesh.cli scale    in.osh out.osh --factor 0.01
    python -m pumiumtally_amd.mesh.cli box      out.osh --cells 55 --extent 1.0

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


if __name__ == "__main__":
    main()
