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
