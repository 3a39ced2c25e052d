"""Mesh utilities: builders, format conversion."""
from .. import Mesh, build_box, mesh_from_arrays, read_gmsh, read_mesh, read_osh


def convert(msh_path: str, osh_dir: str):
    """Convert a Gmsh ASCII .msh file to a .osh mesh directory (the role of
    Omega_h's msh2osh in the reference workflow, README.md:115-126)."""
    m = read_gmsh(msh_path)
    m.write_osh(osh_dir)
    return m


def box_mesh_with_tets(target_tets: int, extent: float = 1.0):
    """Box mesh with ~target_tets elements (6 tets per cell)."""
    n = max(1, round((target_tets / 6.0) ** (1.0 / 3.0)))
    return build_box(n, n, n, extent, extent, extent), n


__all__ = ["Mesh", "build_box", "mesh_from_arrays", "read_gmsh", "read_mesh",
           "read_osh", "convert", "box_mesh_with_tets"]


def write_pvtu(basename, mesh, owners, rank, world, fields):
    """Parallel VTK output: each rank writes a .vtu piece holding ONLY
    its owned elements; rank 0 also writes the .pvtu master referencing
    every piece.  The analog of Omega_h::vtk::write_parallel (the
    reference's writer, PumiTallyImpl.cpp:415) for real decompositions:
    pieces tile the mesh exactly (ghosts excluded), so ParaView shows
    the assembled whole.

    basename: path without extension (writes basename.pvtu +
    basename_p<rank>.vtu).  fields: list of (name, per-GLOBAL-element
    values) -- pass globally-reduced values (e.g. normalized
    flux_global) so every piece carries its elements' true totals.
    Caller is responsible for a barrier afterwards if readers follow.
    """
    import os

    import numpy as np

    owners = np.asarray(owners)
    mine = np.flatnonzero(owners == rank)
    tets = np.asarray(mesh.tet2vert).reshape(-1, 4)[mine]
    used, inv = np.unique(tets.ravel(), return_inverse=True)
    coords = np.asarray(mesh.coords).reshape(-1, 3)[used]
    piece = mesh_from_arrays(np.ascontiguousarray(coords, np.float64),
                             np.ascontiguousarray(
                                 inv.reshape(-1, 4).astype(np.int32)))
    piece_fields = [(name, np.ascontiguousarray(
        np.asarray(vals).reshape(-1)[mine], np.float64))
        for name, vals in fields]
    pdir = os.path.dirname(os.path.abspath(basename))
    pbase = os.path.basename(basename)
    piece_name = f"{pbase}_p{rank}.vtu"
    piece.write_vtk_fields(os.path.join(pdir, piece_name), piece_fields)
    if rank == 0:
        lines = ['<?xml version="1.0"?>',
                 '<VTKFile type="PUnstructuredGrid" version="1.0" '
                 'byte_order="LittleEndian" header_type="UInt64">',
                 '<PUnstructuredGrid GhostLevel="0">',
                 '<PPoints><PDataArray type="Float64" '
                 'NumberOfComponents="3"/></PPoints>',
                 '<PCellData>']
        for name, _ in fields:
            lines.append(f'<PDataArray type="Float64" Name="{name}"/>')
        lines.append('</PCellData>')
        for r in range(world):
            lines.append(f'<Piece Source="{pbase}_p{r}.vtu"/>')
        lines += ['</PUnstructuredGrid>', '</VTKFile>', '']
        with open(f"{basename}.pvtu", "w") as f:
            f.write("\n".join(lines))
