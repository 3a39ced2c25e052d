"""Mesh core tests: topology, adjacency, geometry, IO.

Mirrors the reference's analytic-fixture strategy: the
6-tet unit cube from build_box is the golden mesh, with hand-computed
volumes, adjacency and localization expectations.
"""
import os

import numpy as np
import pytest

import pumiumtally_amd as pt


def test_build_box_unit_cube():
    m = pt.build_box(1, 1, 1)
    assert m.nelems == 6
    assert m.nverts == 8
    # element 0 centroid pinned by the reference tests
    assert np.allclose(m.centroid(0), (0.5, 0.75, 0.25))
    v = m.volumes
    assert np.allclose(v, 1.0 / 6.0)
    assert abs(v.sum() - 1.0) < 1e-14


def test_build_box_adjacency_symmetric():
    m = pt.build_box(3, 2, 4, 3.0, 2.0, 4.0)
    nbr = m.neighbors
    assert m.nelems == 3 * 2 * 4 * 6
    assert np.allclose(m.volumes.sum(), 3.0 * 2.0 * 4.0)
    n_boundary = 0
    for t in range(m.nelems):
        for f in range(4):
            o = nbr[t, f]
            if o == -1:
                n_boundary += 1
            else:
                assert t in nbr[o], f"adjacency not symmetric at tet {t} face {f}"
    # surface of a box cut into 6-tet cells: every boundary quad face of a
    # cell contributes 2 triangles
    expected_boundary = 2 * 2 * (3 * 2 + 2 * 4 + 3 * 4)
    assert n_boundary == expected_boundary


def test_positive_orientation():
    m = pt.build_box(2, 2, 2)
    assert (m.volumes > 0).all()


def test_locate():
    m = pt.build_box(1, 1, 1)
    # coordinate-ordering regions pinned by the reference tests
    pts = np.array([
        [0.1, 0.4, 0.5],   # z>=y>=x -> el 2
        [0.9, 0.4, 0.5],   # x>=z>=y -> el 4
        [0.45, 0.4, 0.5],  # z>=x>=y -> el 3
        [0.5, 0.75, 0.25], # el 0 centroid
    ])
    ids = m.locate(pts)
    assert list(ids[:3]) == [2, 4, 3]
    assert ids[3] == 0
    outside = m.locate(np.array([[1.5, 0.5, 0.5], [-0.1, 0.2, 0.2]]))
    assert list(outside) == [-1, -1]


def test_locate_random_consistency():
    m = pt.build_box(4, 4, 4, 2.0, 2.0, 2.0)
    rng = np.random.default_rng(0)
    pts = rng.uniform(0.001, 1.999, size=(500, 3))
    ids = m.locate(pts)
    assert (ids >= 0).all()
    # each located tet must actually contain its point (barycentric check)
    coords = m.coords
    tets = m.tet2vert
    for p, t in zip(pts, ids):
        vs = coords[tets[t]]
        mat = np.column_stack([vs[1] - vs[0], vs[2] - vs[0], vs[3] - vs[0]])
        bary = np.linalg.solve(mat, p - vs[0])
        assert bary.min() > -1e-9 and bary.sum() < 1 + 1e-9


def test_osh_roundtrip(tmp_path):
    m = pt.build_box(2, 3, 1)
    d = str(tmp_path / "mesh.osh")
    m.write_osh(d)
    m2 = pt.read_osh(d)
    assert m2.nelems == m.nelems
    assert np.array_equal(m2.tet2vert, m.tet2vert)
    assert np.array_equal(m2.coords, m.coords)
    # read_mesh dispatch on extension, with or without trailing slash
    m3 = pt.read_mesh(d)
    assert m3.nelems == m.nelems
    m4 = pt.read_mesh(d + "/")
    assert m4.nelems == m.nelems


def test_gmsh_v2_reader(tmp_path):
    # unit cube corners; single tet mesh written as Gmsh 2.2 ASCII
    msh = tmp_path / "t.msh"
    msh.write_text(
        "$MeshFormat\n2.2 0 8\n$EndMeshFormat\n"
        "$Nodes\n4\n1 0 0 0\n2 1 0 0\n3 0 1 0\n4 0 0 1\n$EndNodes\n"
        "$Elements\n2\n1 2 2 0 1 1 2 3\n2 4 2 0 1 1 2 3 4\n$EndElements\n"
    )
    m = pt.read_gmsh(str(msh))
    assert m.nelems == 1
    assert m.nverts == 4
    assert np.allclose(m.volumes, [1.0 / 6.0])


def test_gmsh_v41_reader(tmp_path):
    msh = tmp_path / "t41.msh"
    msh.write_text(
        "$MeshFormat\n4.1 0 8\n$EndMeshFormat\n"
        "$Nodes\n1 4 1 4\n3 1 0 4\n1\n2\n3\n4\n"
        "0 0 0\n1 0 0\n0 1 0\n0 0 1\n$EndNodes\n"
        "$Elements\n1 1 1 1\n3 1 4 1\n1 1 2 3 4\n$EndElements\n"
    )
    m = pt.read_gmsh(str(msh))
    assert m.nelems == 1
    assert np.allclose(m.volumes, [1.0 / 6.0])


def test_vtk_output(tmp_path):
    m = pt.build_box(1, 1, 1)
    out = str(tmp_path / "flux.vtk")
    pt.write_tally_vtk(out, m, np.arange(6, dtype=float))
    text = open(out).read()
    assert "UNSTRUCTURED_GRID" in text
    assert "SCALARS flux double" in text
    assert "SCALARS volume double" in text
    assert "CELL_DATA 6" in text


def test_vtk_binary_output(tmp_path):
    import struct
    # auto mode: small meshes write ASCII, >200k-element meshes binary
    m = pt.build_box(2, 2, 2)
    out = str(tmp_path / "b.vtk")
    pt.write_tally_vtk(out, m, np.arange(m.nelems, dtype=float))
    head = open(out, "rb").read(200).decode(errors="ignore")
    assert "ASCII" in head  # small mesh -> ascii
    big = pt.build_box(33, 33, 33)  # 215k tets -> auto binary
    out2 = str(tmp_path / "big.vtk")
    pt.write_tally_vtk(out2, big, np.ones(big.nelems))
    data = open(out2, "rb").read()
    assert b"BINARY" in data[:200]
    # decode the first point (big-endian doubles) and check it's the origin
    i = data.index(b"POINTS")
    j = data.index(b"\n", i) + 1
    x, y, z = struct.unpack(">3d", data[j:j + 24])
    assert (x, y, z) == (0.0, 0.0, 0.0)
    # file is much smaller than an ASCII equivalent would be
    assert len(data) < 40_000_000


def test_degenerate_element_rejected():
    coords = np.array([[0, 0, 0], [1, 0, 0], [0, 1, 0], [0, 0, 1]], float)
    tets = np.array([[0, 1, 2, 2]], np.int32)  # repeated vertex -> zero volume
    with pytest.raises(RuntimeError):
        pt.mesh_from_arrays(coords, tets)


def test_nonmanifold_mesh_rejected():
    # three tets sharing the (0,1,2) face
    coords = np.array([[0, 0, 0], [1, 0, 0], [0, 1, 0],
                       [0, 0, 1], [0, 0, -1], [1, 1, 1]], float)
    tets = np.array([[0, 1, 2, 3], [0, 1, 2, 4], [0, 1, 2, 5]], np.int32)
    with pytest.raises(RuntimeError):
        pt.mesh_from_arrays(coords, tets)


def test_gmsh_binary_rejected(tmp_path):
    msh = tmp_path / "bin.msh"
    msh.write_text("$MeshFormat\n2.2 1 8\n$EndMeshFormat\n")
    with pytest.raises(RuntimeError):
        pt.read_gmsh(str(msh))


def test_osh_foreign_magic_rejected(tmp_path):
    d = tmp_path / "foreign.osh"
    d.mkdir()
    (d / "0.osh").write_bytes(b"\xa1\x1a" + b"\x00" * 64)  # Omega_h-style magic
    with pytest.raises(RuntimeError, match="Omega_h binary mesh"):
        pt.read_osh(str(d))


def test_vtu_writer_roundtrip(tmp_path):
    """XML .vtu output: self-parse the raw appended blocks and verify the
    geometry and cell fields round-trip bitwise."""
    m = pt.build_box(2, 2, 2)
    n = 30
    rng = np.random.default_rng(6)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = rng.uniform(0.1, 0.9, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    eng = pt.TallyEngine(m, n, device="cpu")
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w)
    out = tmp_path / "tally.vtu"
    eng.write_tally_results(str(out))

    raw = out.read_bytes()
    header = raw.split(b"<AppendedData", 1)[0].decode()
    assert 'type="UnstructuredGrid"' in header
    assert 'Name="flux"' in header and 'Name="volume"' in header
    payload = raw.split(b'encoding="raw">_', 1)[1]

    def block(off):
        size = int.from_bytes(payload[off:off + 8], "little")
        return payload[off + 8:off + 8 + size], off + 8 + size

    import re
    offsets = [int(x) for x in re.findall(r'offset="(\d+)"', header)]
    pts, _ = block(offsets[0])
    conn, _ = block(offsets[1])
    offs, _ = block(offsets[2])
    types, _ = block(offsets[3])
    flux, _ = block(offsets[4])
    vol, _ = block(offsets[5])

    assert np.frombuffer(pts, np.float64).size == m.nverts * 3
    c = np.frombuffer(conn, np.int64).reshape(-1, 4)
    assert c.shape[0] == m.nelems and c.max() == m.nverts - 1
    assert np.frombuffer(offs, np.int64)[-1] == m.nelems * 4
    assert set(np.frombuffer(types, np.uint8)) == {10}
    expected = pt.normalize_flux(m, eng.flux())
    assert np.array_equal(np.frombuffer(flux, np.float64), expected)
    assert np.allclose(np.frombuffer(vol, np.float64), np.asarray(m.volumes))


def test_vtu_grouped_fields(tmp_path):
    m = pt.build_box(2, 2, 2)
    n, G = 30, 2
    rng = np.random.default_rng(7)
    o = rng.uniform(0.1, 0.9, size=(n, 3))
    d = rng.uniform(0.1, 0.9, size=(n, 3))
    w = rng.uniform(0.1, 1.0, n)
    g = rng.integers(0, G, n).astype(np.uint16)
    eng = pt.TallyEngine(m, n, device="cpu", ngroups=G)
    eng.copy_initial_position(o.ravel())
    eng.move(o.ravel(), d.ravel(), np.ones(n, np.int8), w, groups=g)
    out = tmp_path / "grouped.vtu"
    eng.write_tally_results(str(out))
    head = out.read_bytes().split(b"<AppendedData", 1)[0].decode()
    assert 'Name="flux"' in head
    assert 'Name="flux_g0"' in head and 'Name="flux_g1"' in head


def test_cli_partition(tmp_path):
    """Mesh CLI partition preview: counts, weighted split, owner export."""
    import subprocess
    import sys

    osh = tmp_path / "m.osh"
    m = pt.build_box(4, 4, 4)
    m.write_osh(str(osh))
    w = np.linspace(1.0, 10.0, m.nelems)
    wpath = tmp_path / "w.npy"
    np.save(wpath, w)
    out = tmp_path / "owners.npy"
    r = subprocess.run(
        [sys.executable, "-m", "pumiumtally_amd.mesh.cli", "partition",
         str(osh), "--parts", "4", "--weights", str(wpath),
         "--out", str(out)],
        capture_output=True, text=True, timeout=120,
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    assert r.returncode == 0, r.stderr
    assert "work" in r.stdout and "cut" in r.stdout
    owners = np.load(out)
    assert owners.shape == (m.nelems,)
    sums = np.array([w[owners == p].sum() for p in range(4)])
    assert sums.max() / sums.min() < 1.3
